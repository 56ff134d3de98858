// Grouped GEMM for dropless MoE experts on MI355X (gfx950).
//
// Replaces the per-expert hipBLASLt loop + .tolist() host sync in the
// round-1 ExpertMLPs (reference contract: NxD ExpertMLPs dropless mode,
// transformer.py:423-464). Tokens arrive sorted by local expert and
// scattered into BM-aligned padded segments (device-side index math in
// ops/moe_gemm.py — nothing syncs to the host):
//
//   fwd   : C[t, n] = sum_k A[t, k]  * W[e(t), n, k]   (x @ W^T, F.linear)
//   dgrad : C[t, k] = sum_n A[t, n]  * W[e(t), n, k]   (dy @ W)
//   wgrad : dW[e, n, k] = sum_{t in seg e} dy[t, n] * x[t, k]
//
// e(t) = row tile t/BM's expert, from a device tile->expert map. Padded
// rows are zero in A (and dy), so they contribute nothing to wgrad and
// their C rows are garbage that the gather-back never reads.
//
// Tiling (the flash-v2 idioms, attn_common.h): 512 threads = 8 waves,
// each wave owns 32 output rows (2 sub-blocks of 16) x BN cols;
// 16x16x32 bf16 MFMA; operands staged through LDS per BK chunk.
#include "attn_common.h"

// ============================ fwd / dgrad ============================
// TRANS_B = false: contraction over W's LAST dim (fwd, x @ W^T):
//   B-frag col n, k-window contiguous -> W rows staged row-major.
// TRANS_B = true: contraction over W's FIRST dim (dgrad, dy @ W):
//   B-frag col k, contraction n -> W chunk staged TRANSPOSED (tr_swz).
template <bool TRANS_B>
__global__ __launch_bounds__(512) void moe_gemm_kernel(
    const bf16* __restrict__ A,   // [Tp, K] row-major (padded, sorted)
    const bf16* __restrict__ W,   // [E, N, Kw] row-major
    bf16* __restrict__ C,         // [Tp, Nc] row-major
    const int* __restrict__ tile_expert,  // [row_tiles] tile -> expert
    const int* __restrict__ total_rows,   // [1] actual padded total
    int K,    // contraction length (fwd: Kw, dgrad: N)
    int Nc,   // C columns (fwd: N, dgrad: Kw)
    long wstride) {  // elements per expert slab (N * Kw)
  constexpr int BM = 256, BN = 128, BK = 32;
  constexpr int AP = BK + 8;
  // B tile: row-major [BN n-rows][BK k] (fwd) or transposed image
  // [BN k-rows][BK n] (dgrad) — both BN rows x BP pitch
  constexpr int BP = BK + 8;
  __shared__ __bf16 a_lds[BM * AP];
  __shared__ __bf16 b_lds[BN * BP];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int rtile = blockIdx.x;
  const int ctile = blockIdx.y;
  if (rtile * BM >= *total_rows) return;
  const int e = tile_expert[rtile];

  const bf16* Ap = A + (long)rtile * BM * K;
  const bf16* We = W + (long)e * wstride;
  const int row_w = wid * 32;  // wave's first row within the tile

  constexpr int DN = BN / 16;
  f32x4_t acc[2][DN];
#pragma unroll
  for (int sb = 0; sb < 2; ++sb)
#pragma unroll
    for (int nj = 0; nj < DN; ++nj) acc[sb][nj] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const int nkb = (K + BK - 1) / BK;
  for (int kb = 0; kb < nkb; ++kb) {
    const int k0 = kb * BK;
    {  // stage A [BM x BK] row-major (BM*BK/8 = 1024 int4 loads, 2/thread)
      constexpr int AV = BM * BK / 8;
      for (int t = threadIdx.x; t < AV; t += 512) {
        const int row = t / (BK / 8);
        const int col8 = (t % (BK / 8)) * 8;
        // K is a multiple of 8 but not necessarily of BK; guard the tail
        int4 v = (k0 + col8 < K)
                     ? *(const int4*)(Ap + (long)row * K + k0 + col8)
                     : int4{0, 0, 0, 0};
        *(int4*)&a_lds[row * AP + col8] = v;
      }
      if (!TRANS_B) {
        // fwd: B rows = W[n] for n in [ctile*BN, +BN), k-window k0
        constexpr int BV = BN * BK / 8;
        for (int t = threadIdx.x; t < BV; t += 512) {
          const int row = t / (BK / 8);
          const int col8 = (t % (BK / 8)) * 8;
          int4 v = (k0 + col8 < K)
                       ? *(const int4*)(We + (long)(ctile * BN + row) * K +
                                        k0 + col8)
                       : int4{0, 0, 0, 0};
          *(int4*)&b_lds[row * BP + col8] = v;
        }
      } else {
        // dgrad: stage W chunk [BK n-rows x BN k-cols] TRANSPOSED so the
        // B-frag contraction (over n) reads contiguously; paired writes
        constexpr int TV = BK * BN / 16;  // row pairs
        for (int t = threadIdx.x; t < TV; t += 512) {
          const int row = (t / (BN / 8)) * 2;   // n-offset within chunk
          const int col8 = (t % (BN / 8)) * 8;  // k-offset within tile
          const int n0 = k0 + row;
          const long base = (long)ctile * BN + col8;
          int4 w0 = (n0 < K) ? *(const int4*)(We + (long)n0 * Nc + base)
                             : int4{0, 0, 0, 0};
          int4 w1 = (n0 + 1 < K)
                        ? *(const int4*)(We + (long)(n0 + 1) * Nc + base)
                        : int4{0, 0, 0, 0};
          const __bf16* e0 = (const __bf16*)&w0;
          const __bf16* e1 = (const __bf16*)&w1;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            __bf16 pr[2] = {e0[j], e1[j]};
            const int r = col8 + j;  // k index = image row
            *(uint*)((char*)b_lds + tr_swz((uint)(r * BP + row) * 2, r)) =
                *(uint*)pr;
          }
        }
      }
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8_t a0 = load_frag_a(&a_lds[row_w * AP], AP, kk * 32, lane);
      bf16x8_t a1 = load_frag_a(&a_lds[(row_w + 16) * AP], AP, kk * 32, lane);
#pragma unroll
      for (int nj = 0; nj < DN; ++nj) {
        bf16x8_t bf;
        if (!TRANS_B)
          bf = load_frag_b_rowmajorT(&b_lds[nj * 16 * BP], BP, kk * 32, lane);
        else
          bf = load_frag_b_trT_swz(b_lds, BP, nj * 16, kk * 32, lane);
        acc[0][nj] = MFMA_16x16x32(a0, bf, acc[0][nj]);
        acc[1][nj] = MFMA_16x16x32(a1, bf, acc[1][nj]);
      }
    }
    __syncthreads();
  }

  // store C tile
  bf16* Cp = C + (long)rtile * BM * Nc + (long)ctile * BN;
#pragma unroll
  for (int sb = 0; sb < 2; ++sb)
#pragma unroll
    for (int nj = 0; nj < DN; ++nj)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = row_w + sb * 16 + (lane >> 4) * 4 + r;
        Cp[(long)row * Nc + nj * 16 + (lane & 15)] = f2bf(acc[sb][nj][r]);
      }
}

// ============================ wgrad ============================
// dW[e, n, k] = sum_t dy[t, n] * x[t, k] over expert e's padded segment.
// Both operands need the contraction (t) contiguous per fragment ->
// both staged transposed. Output fp32 (optimizer-side grads are fp32).
__global__ __launch_bounds__(512) void moe_wgrad_kernel(
    const bf16* __restrict__ dY,  // [Tp, N]
    const bf16* __restrict__ X,   // [Tp, K]
    float* __restrict__ dW,       // [E, N, K] fp32
    const int* __restrict__ seg_start,  // [E+1] padded segment bounds
    int N, int K) {
  constexpr int BM = 128, BN = 128, BT = 32;  // n-tile, k-tile, t-chunk
  // transposed images [dim][t]: dim-major rows, t contiguous per row
  constexpr int IP = BT + 8;
  __shared__ __bf16 dy_img[BM * IP];
  __shared__ __bf16 x_img[BN * IP];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int e = blockIdx.z;
  const int ntile = blockIdx.x;  // over N/BM
  const int ktile = blockIdx.y;  // over K/BN
  const int t0 = seg_start[e], t1 = seg_start[e + 1];

  // 8 waves cover the 128x128 tile as 4 row-groups x 2 col-halves
  const int rw = (wid & 3) * 32;          // n-offset (0..96)
  const int cw = (wid >> 2) * (BN / 2);   // k-offset half (0 or 64)
  constexpr int DNH = (BN / 2) / 16;      // 4 col frags per wave

  f32x4_t acch[2][DNH];
#pragma unroll
  for (int sb = 0; sb < 2; ++sb)
#pragma unroll
    for (int nj = 0; nj < DNH; ++nj)
      acch[sb][nj] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  for (int tb = t0; tb < t1; tb += BT) {
    {  // stage dy^T [BM n x BT t] and x^T [BN k x BT t] images (paired)
      constexpr int PV = BM * BT / 16;  // row pairs per image
      for (int t = threadIdx.x; t < PV; t += 512) {
        const int trow = (t / (BM / 8)) * 2;   // t-offset within chunk
        const int col8 = (t % (BM / 8)) * 8;   // n-offset
        const long g0 = tb + trow, g1 = g0 + 1;
        int4 d0 = *(const int4*)(dY + g0 * N + (long)ntile * BM + col8);
        int4 d1 = *(const int4*)(dY + g1 * N + (long)ntile * BM + col8);
        int4 x0 = *(const int4*)(X + g0 * K + (long)ktile * BN + col8);
        int4 x1 = *(const int4*)(X + g1 * K + (long)ktile * BN + col8);
        const __bf16 *de0 = (const __bf16*)&d0, *de1 = (const __bf16*)&d1;
        const __bf16 *xe0 = (const __bf16*)&x0, *xe1 = (const __bf16*)&x1;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          __bf16 dp[2] = {de0[j], de1[j]};
          __bf16 xp[2] = {xe0[j], xe1[j]};
          const int r = col8 + j;  // image row = n (or k)
          const uint byte = tr_swz((uint)(r * IP + trow) * 2, r);
          *(uint*)((char*)dy_img + byte) = *(uint*)dp;
          *(uint*)((char*)x_img + byte) = *(uint*)xp;
        }
      }
    }
    __syncthreads();

    // A-frag: rows = n (from dy image), contraction t (32 = one chunk)
    bf16x8_t a0 = load_frag_b_trT_swz(dy_img, IP, rw, 0, lane);
    bf16x8_t a1 = load_frag_b_trT_swz(dy_img, IP, rw + 16, 0, lane);
#pragma unroll
    for (int nj = 0; nj < DNH; ++nj) {
      bf16x8_t bf = load_frag_b_trT_swz(x_img, IP, cw + nj * 16, 0, lane);
      acch[0][nj] = MFMA_16x16x32(a0, bf, acch[0][nj]);
      acch[1][nj] = MFMA_16x16x32(a1, bf, acch[1][nj]);
    }
    __syncthreads();
  }

  float* dWp = dW + (long)e * N * K + (long)ntile * BM * K + (long)ktile * BN;
#pragma unroll
  for (int sb = 0; sb < 2; ++sb)
#pragma unroll
    for (int nj = 0; nj < DNH; ++nj)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int n = rw + sb * 16 + (lane >> 4) * 4 + r;
        dWp[(long)n * K + cw + nj * 16 + (lane & 15)] = acch[sb][nj][r];
      }
}

extern "C" {
void launch_moe_gemm(const void* a, const void* w, void* c,
                     const int* tile_expert, const int* total_rows,
                     int row_tiles, int K, int Nc, long wstride, bool trans_b,
                     hipStream_t stream) {
  dim3 grid(row_tiles, Nc / 128);
  dim3 blk(512);
  if (trans_b)
    moe_gemm_kernel<true><<<grid, blk, 0, stream>>>(
        (const bf16*)a, (const bf16*)w, (bf16*)c, tile_expert, total_rows, K,
        Nc, wstride);
  else
    moe_gemm_kernel<false><<<grid, blk, 0, stream>>>(
        (const bf16*)a, (const bf16*)w, (bf16*)c, tile_expert, total_rows, K,
        Nc, wstride);
}

void launch_moe_wgrad(const void* dy, const void* x, void* dw,
                      const int* seg_start, int E, int N, int K,
                      hipStream_t stream) {
  dim3 grid(N / 128, K / 128, E);
  dim3 blk(512);
  moe_wgrad_kernel<<<grid, blk, 0, stream>>>((const bf16*)dy, (const bf16*)x,
                                             (float*)dw, seg_start, N, K);
}
}
