// Causal flash attention FORWARD for MI355X (gfx950).
// MFMA 16×16×32 bf16 tiles, LDS-staged K/V, online softmax in registers.
// Replaces the reference's NKI flash kernel contract (SURVEY.md §2.3:
// nki_flash_attn_func, modeling_llama.py:486): causal, bf16, GQA by
// kv-head indexing, per-row LSE saved for the recompute backward.
//
// Geometry: block = 4 waves (256 thr); BM = 64 query rows (16/wave),
// BN = 64 keys per tile; D = head_dim (64 or 128, template).
// LDS: K[64][D+8] + V^T[D][64+8] + P[4][16][64+8], bf16.
#include "attn_common.h"

template <int D, bool CAUSAL>
__global__ __launch_bounds__(256) void flash_fwd_kernel(
    const bf16* __restrict__ Q,  // [B, HQ, S, D]
    const bf16* __restrict__ K,  // [B, HKV, S, D]
    const bf16* __restrict__ V,  // [B, HKV, S, D]
    bf16* __restrict__ O,        // [B, HQ, S, D]
    float* __restrict__ LSE,     // [B, HQ, S]
    int S, int HQ, int HKV, float scale) {
  constexpr int BM = 64, BN = 64;
  constexpr int KP = D + 8;   // padded row stride for K rows (bank spread)
  constexpr int VP = BN + 8;  // padded row stride for V^T / P rows
  __shared__ __bf16 k_lds[BN * KP];
  __shared__ __bf16 vt_lds[D * VP];
  __shared__ __bf16 p_lds[4 * 16 * VP];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int qblock = blockIdx.x;
  const int bh = blockIdx.y;  // b * HQ + hq
  const int hq = bh % HQ;
  const int b = bh / HQ;
  const int hkv = hq / (HQ / HKV);

  const long qoff = ((long)b * HQ + hq) * S * D;
  const long kvoff = ((long)b * HKV + hkv) * S * D;
  const bf16* Qp = Q + qoff;
  const bf16* Kp = K + kvoff;
  const bf16* Vp = V + kvoff;

  const int q0 = qblock * BM;           // first query row of this block
  const int qrow_w = q0 + wid * 16;     // first row of this wave

  // ---- Q fragments: 16 rows × D, registers (A-frags) ----
  constexpr int DK = D / 32;  // k-chunks of 32
  bf16x8_t qfrag[DK];
  {
    const int r = qrow_w + (lane & 15);
    const int row = (r < S) ? r : (S - 1);  // clamp; masked rows unused
#pragma unroll
    for (int kk = 0; kk < DK; ++kk) {
      const bf16* p = Qp + (long)row * D + kk * 32 + (lane >> 4) * 8;
      *(int4*)&qfrag[kk] = *(const int4*)p;
    }
  }

  // online-softmax state: this lane covers rows (lane>>4)*4 + r (r=0..3)
  float m_i[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float l_i[4] = {0.f, 0.f, 0.f, 0.f};
  constexpr int DN = D / 16;  // output col chunks
  f32x4_t oacc[DN];
#pragma unroll
  for (int nj = 0; nj < DN; ++nj) oacc[nj] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const int kend = CAUSAL ? min(S, q0 + BM) : S;
  const int nkb = (kend + BN - 1) / BN;

  for (int jb = 0; jb < nkb; ++jb) {
    const int kbase = jb * BN;
    // ---- stage K tile [BN][D] and V^T tile [D][BN] ----
    {
      // 256 threads; K rows: each thread loads (BN*D/8)/256 int4s
      constexpr int VECS = BN * D / 8;  // int4 count
      for (int t = threadIdx.x; t < VECS; t += 256) {
        const int row = t / (D / 8);
        const int col8 = (t % (D / 8)) * 8;
        const int gr = kbase + row;
        int4 val;
        if (gr < S)
          val = *(const int4*)(Kp + (long)gr * D + col8);
        else
          val = int4{0, 0, 0, 0};
        *(int4*)&k_lds[row * KP + col8] = val;
        // V: load same pattern, store transposed
        int4 vv;
        if (gr < S)
          vv = *(const int4*)(Vp + (long)gr * D + col8);
        else
          vv = int4{0, 0, 0, 0};
        const __bf16* ve = (const __bf16*)&vv;
#pragma unroll
        for (int j = 0; j < 8; ++j) vt_lds[(col8 + j) * VP + row] = ve[j];
      }
    }
    __syncthreads();

    // ---- S = Q K^T (scaled): 4 key chunks of 16 ----
    f32x4_t sacc[4];
#pragma unroll
    for (int nk = 0; nk < 4; ++nk) {
      sacc[nk] = f32x4_t{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < DK; ++kk) {
        bf16x8_t bfrag =
            load_frag_b_rowmajorT(&k_lds[nk * 16 * KP], KP, kk * 32, lane);
        sacc[nk] = MFMA_16x16x32(qfrag[kk], bfrag, sacc[nk]);
      }
    }

    // ---- mask + online softmax ----
    float tile_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
    float sv[4][4];
#pragma unroll
    for (int nk = 0; nk < 4; ++nk) {
      const int kcol = kbase + nk * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = qrow_w + (lane >> 4) * 4 + r;
        float s = sacc[nk][r] * scale;
        const bool dead = (kcol >= S) || (CAUSAL && kcol > qrow);
        s = dead ? -1e30f : s;
        sv[nk][r] = s;
        tile_max[r] = fmaxf(tile_max[r], s);
      }
    }
    // row-reduce max over the 16 lanes of the key group
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        tile_max[r] = fmaxf(tile_max[r], __shfl_xor(tile_max[r], off, 64));
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float mn = fmaxf(m_i[r], tile_max[r]);
      alpha[r] = __expf(m_i[r] - mn);
      m_i[r] = mn;
    }
    // P = exp(s - m), row sums, write P to LDS (bf16)
    float rsum[4] = {0.f, 0.f, 0.f, 0.f};
    __bf16* pw = &p_lds[wid * 16 * VP];
#pragma unroll
    for (int nk = 0; nk < 4; ++nk) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = __expf(sv[nk][r] - m_i[r]);
        rsum[r] += p;
        pw[((lane >> 4) * 4 + r) * VP + nk * 16 + (lane & 15)] = (__bf16)p;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        rsum[r] += __shfl_xor(rsum[r], off, 64);
      l_i[r] = l_i[r] * alpha[r] + rsum[r];
    }
    // rescale O
#pragma unroll
    for (int nj = 0; nj < DN; ++nj) {
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[nj][r] *= alpha[r];
    }
    __syncthreads();  // P visible to own wave only — but keep tile barrier
                      // aligned for the next K/V stage overwrite

    // ---- O += P V : A = P (16×BN), B = V (BN×D) via V^T LDS ----
#pragma unroll
    for (int nj = 0; nj < DN; ++nj) {
#pragma unroll
      for (int kk = 0; kk < BN / 32; ++kk) {
        bf16x8_t pa = load_frag_a(pw, VP, kk * 32, lane);
        bf16x8_t vb =
            load_frag_b_rowmajorT(&vt_lds[nj * 16 * VP], VP, kk * 32, lane);
        oacc[nj] = MFMA_16x16x32(pa, vb, oacc[nj]);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: O /= l, store O + LSE ----
  float inv_l[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) inv_l[r] = (l_i[r] > 0.f) ? 1.f / l_i[r] : 0.f;
  bf16* Op = O + qoff;
#pragma unroll
  for (int nj = 0; nj < DN; ++nj) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = qrow_w + (lane >> 4) * 4 + r;
      if (qrow < S)
        Op[(long)qrow * D + nj * 16 + (lane & 15)] =
            f2bf(oacc[nj][r] * inv_l[r]);
    }
  }
  if ((lane & 15) == 0) {
    float* Lp = LSE + ((long)b * HQ + hq) * S;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = qrow_w + (lane >> 4) * 4 + r;
      if (qrow < S) Lp[qrow] = m_i[r] + __logf(fmaxf(l_i[r], 1e-30f));
    }
  }
}

extern "C" {
void launch_flash_fwd(const void* q, const void* k, const void* v, void* o,
                      void* lse, int B, int HQ, int HKV, int S, int D,
                      bool causal, float scale, hipStream_t stream) {
  dim3 grid((S + 63) / 64, B * HQ);
  dim3 blk(256);
#define CASE(DD, CC)                                                        \
  flash_fwd_kernel<DD, CC><<<grid, blk, 0, stream>>>(                       \
      (const bf16*)q, (const bf16*)k, (const bf16*)v, (bf16*)o, (float*)lse, \
      S, HQ, HKV, scale)
  if (D == 128) {
    if (causal) CASE(128, true); else CASE(128, false);
  } else if (D == 64) {
    if (causal) CASE(64, true); else CASE(64, false);
  }
#undef CASE
}
}
