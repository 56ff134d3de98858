#include "hip/hip_runtime.h"
// Flash attention forward v3 — T12 swapped-operand structure (DARK: only
// dispatched when NXDT_ATTN_V3=1; numerics to be confirmed on HW, see
// ROADMAP.md §1. The 32×32×16 fragment maps are probe-verified:
// tests/test_gpu_kernels.py::test_mfma32_layout_probe).
//
// Design (guide §5.5 T12/T16): compute S^T = K·Q^T with 32×32×16 MFMAs so
// each lane owns ONE query column (q = lane&31; the two wave halves hold
// complementary key rows). Softmax state (m, l) is lane-SCALAR; P never
// touches LDS: it is packed to bf16 with v_cvt_pk (compiler form) and the
// halves are exchanged with permlane32_swap to assemble the PV B-fragments
// in registers. O^T accumulates per lane (one q column × d rows); the
// epilogue transposes through LDS once for coalesced stores.
//
// Geometry: 8 waves × 32 queries = BM 256; BN = 32 keys/tile staged in
// LDS (K row-major + V^T with the tr_swz XOR swizzle).
#include "attn_common.h"

typedef float f32x16_t __attribute__((ext_vector_type(16)));

#define MFMA_32x32x16(A, B, C) \
  __builtin_amdgcn_mfma_f32_32x32x16_bf16((A), (B), (C), 0, 0, 0)

template <int D, bool CAUSAL>
__global__ __launch_bounds__(512) void flash_fwd_v3_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, bf16* __restrict__ O,
    float* __restrict__ LSE, int S, int Bb, int HQ, int HKV, float scale,
    int window, long sQs, long sQb, long sQh, long sKs, long sKb, long sKh,
    long sVs, long sVb, long sVh) {
  constexpr int BM = 256, BN = 32;
  constexpr int KP = D + 8;   // K image row stride
  constexpr int VP = BN + 8;  // V^T image row stride (40 elems, 80 B)
  constexpr int OP = D + 8;   // epilogue O image row stride
  // one __shared__ object: staging while looping, transpose at epilogue
  __shared__ __bf16 smem[(BN * KP + D * VP) > (8 * 32 * OP)
                             ? (BN * KP + D * VP)
                             : (8 * 32 * OP)];
  __bf16* k_lds = smem;
  __bf16* vt_lds = smem + BN * KP;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int hi = lane >> 5;     // wave half
  const int qcol = lane & 31;   // this lane's query within the wave tile
  const int qblock = blockIdx.x;
  const int bh = blockIdx.y;
  const int hq = bh % HQ;
  const int b = bh / HQ;
  const int hkv = hq / (HQ / HKV);

  const bf16* Qp = Q + b * sQb + hq * sQh;
  const bf16* Kp = K + b * sKb + hkv * sKh;
  const bf16* Vp = V + b * sVb + hkv * sVh;

  const int q0 = qblock * BM;
  const int qrow_w = q0 + wid * 32;
  const int my_q = qrow_w + qcol;  // this lane's (global) query row

  // ---- Q B-fragments (for S^T = K·Q^T): B[k=d][j=q] ← Q[q][d],
  // 8 contiguous d per lane; chunk ck covers d ∈ [16ck, 16ck+16) ----
  constexpr int CK = D / 16;  // 8 for D=128
  bf16x8_t qb[CK];
  {
    const long row = (my_q < S) ? my_q : (S - 1);
#pragma unroll
    for (int ck = 0; ck < CK; ++ck) {
      const bf16* p = Qp + row * sQs + ck * 16 + hi * 8;
      *(int4*)&qb[ck] = *(const int4*)p;
      // fold softmax scale into Q once
#pragma unroll
      for (int j = 0; j < 8; ++j)
        qb[ck][j] = (__bf16)((float)qb[ck][j] * scale);
    }
  }

  float m_i = -1e30f, l_i = 0.f;  // lane-scalar online-softmax state
  constexpr int DC = D / 32;      // O^T d-chunks of 32 rows
  f32x16_t oacc[DC];
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) oacc[dc] = f32x16_t{};

  const int kend = CAUSAL ? min(S, q0 + BM) : S;
  const int nkb = (kend + BN - 1) / BN;
  const int jb0 =
      (CAUSAL && window > 0) ? max(0, (q0 - window + 1) / BN) : 0;
  const int wrow_max = qrow_w + 31;

  for (int jb = jb0; jb < nkb; ++jb) {
    const int kbase = jb * BN;
    // ---- stage K [BN][D] + V^T [D][BN] (tr_swz) ----
    {
      constexpr int KVECS = BN * D / 8;  // 512 int4s for D=128
      const int t = threadIdx.x;
      {
        const int row = t / (D / 8);
        const int col8 = (t % (D / 8)) * 8;
        const int gr = kbase + row;
        int4 kv = (gr < S) ? *(const int4*)(Kp + (long)gr * sKs + col8)
                           : int4{0, 0, 0, 0};
        *(int4*)&k_lds[row * KP + col8] = kv;
      }
      {
        // V pairs: 256 row-pair slices over 512 threads → half the
        // threads idle here at D=128; acceptable for v3.0
        const int tp2 = t;
        if (tp2 < BN * D / 16) {
          const int row = (tp2 / (D / 8)) * 2;
          const int col8 = (tp2 % (D / 8)) * 8;
          const int g0 = kbase + row, g1 = g0 + 1;
          int4 v0 = (g0 < S) ? *(const int4*)(Vp + (long)g0 * sVs + col8)
                             : int4{0, 0, 0, 0};
          int4 v1 = (g1 < S) ? *(const int4*)(Vp + (long)g1 * sVs + col8)
                             : int4{0, 0, 0, 0};
          const __bf16* e0 = (const __bf16*)&v0;
          const __bf16* e1 = (const __bf16*)&v1;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            __bf16 pr[2] = {e0[j], e1[j]};
            const int r = col8 + j;
            *(uint*)((char*)vt_lds + tr_swz((uint)(r * VP + row) * 2, r)) =
                *(uint*)pr;
          }
        }
      }
    }
    __syncthreads();

    if (!CAUSAL || kbase <= wrow_max) {
      // ---- S^T = K · Q^T : A = K rows, B = qb ----
      f32x16_t st = f32x16_t{};
#pragma unroll
      for (int ck = 0; ck < CK; ++ck) {
        bf16x8_t ka;
        // A[row=key=lane&31][k=d=16ck + hi*8 + j]
        *(int4*)&ka = *(const int4*)&k_lds[(lane & 31) * KP + ck * 16 + hi * 8];
        st = MFMA_32x32x16(ka, qb[ck], st);
      }
      // key row of register r: (r&3) + 8*(r>>2) + 4*hi  (+ kbase)
      const bool full_tile =
          (kbase + BN <= S) && (!CAUSAL || (kbase + BN - 1 <= qrow_w)) &&
          (window <= 0 || kbase >= qrow_w + 31 - window + 1);
      float sv[16];
      float tmax = -1e30f;
      if (full_tile) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          sv[r] = st[r];
          tmax = fmaxf(tmax, sv[r]);
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int krow = kbase + (r & 3) + 8 * (r >> 2) + 4 * hi;
          bool dead = (krow >= S) || (CAUSAL && krow > my_q);
          if (CAUSAL && window > 0) dead |= (krow <= my_q - window);
          sv[r] = dead ? -1e30f : st[r];
          tmax = fmaxf(tmax, sv[r]);
        }
      }
      // combine with the partner half-lane (same query)
      tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
      // exact skip when no growth anywhere in the wave
      float alpha = 1.0f;
      if (__builtin_amdgcn_ballot_w64(tmax > m_i) != 0ull) {
        const float mn = fmaxf(m_i, tmax);
        alpha = __expf(m_i - mn);
        m_i = mn;
#pragma unroll
        for (int dc = 0; dc < DC; ++dc)
#pragma unroll
          for (int r = 0; r < 16; ++r) oacc[dc][r] *= alpha;
      }
      // P = exp(s − m) and row-sum (this lane's 16 keys + partner's)
      float p[16];
      float rsum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = __expf(sv[r] - m_i);
        rsum += p[r];
      }
      rsum += __shfl_xor(rsum, 32, 64);
      l_i = l_i * alpha + rsum;

      // ---- pack P to bf16 pairs and build the PV B-fragments in
      // registers via permlane32_swap (T12): reg r holds key
      // (r&3)+8*(r>>2)+4*hi; pack (r, r+1) pairs → u32 of keys
      // {4k+4hi, 4k+4hi+1} etc. ----
      uint pk[8];
#pragma unroll
      for (int g = 0; g < 8; ++g) {
        __bf16 pr[2] = {(__bf16)p[2 * g], (__bf16)p[2 * g + 1]};
        pk[g] = *(uint*)pr;
      }
      // pk layout per half: g = 2*(keygroup) + (pairlow?) with
      // keygroup = (g>>1) → keys 8*(g>>1) + 4*hi + 2*(g&1) + {0,1}
      // swap partner pairs so every lane holds keys 8*(g>>1)+{0..7}:
      //   frag slot order per 8-key window w (= g>>1):
      //     hi=0 lane: [pk0 pk1 | partner pk0 pk1]
      //     hi=1 lane: [partner pk0 pk1 | pk0 pk1]
      // Derivation (keys per reg: (r&3) + 8*(r>>2) + 4*hi):
      //   A0 = pk[4c+0] holds keys 16c + 4hi + {0,1}; A1: +{2,3};
      //   B0 = pk[4c+2] holds keys 16c + 8 + 4hi + {0,1}; B1: +{2,3}.
      // The B-frag for chunk c wants, per lane, keys 16c + 8·hi + {0..7}
      // as 4 u32 pairs. permlane32_swap(x, y) → (x with its upper-half
      // lanes replaced by y's lower half, y with its lower half replaced
      // by x's upper half), and partner lanes share the same query, so:
      //   r0 = swap(A0, B0): slot0 = r0[0], slot2 = r0[1]
      //   r1 = swap(A1, B1): slot1 = r1[0], slot3 = r1[1]
      bf16x8_t pb[2];  // PV B-frags for key chunks [0..15], [16..31]
#pragma unroll
      for (int half = 0; half < 2; ++half) {  // key chunk of 16
        uint a0 = pk[4 * half + 0], a1 = pk[4 * half + 1];
        uint b0 = pk[4 * half + 2], b1 = pk[4 * half + 3];
        auto r0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
        auto r1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
        uint frag[4] = {(uint)r0[0], (uint)r1[0], (uint)r0[1], (uint)r1[1]};
        pb[half] = *(bf16x8_t*)frag;
      }

      // ---- O^T += V^T · P^T ----
#pragma unroll
      for (int dc = 0; dc < DC; ++dc) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          bf16x8_t va;  // A[row=d=dc*32+(lane&31)][k=key=kk*16+hi*8+j]
          const int drow = dc * 32 + (lane & 31);
          const uint byte =
              tr_swz((uint)(drow * VP + kk * 16 + hi * 8) * 2, drow);
          *(int4*)&va = *(const int4*)((const char*)vt_lds + byte);
          oacc[dc] = MFMA_32x32x16(va, pb[kk], oacc[dc]);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: transpose O^T through LDS, coalesced stores ----
  const float inv_l = (l_i > 0.f) ? 1.f / l_i : 0.f;
  __bf16* ow = smem + wid * 32 * OP;  // per-wave [32 q][D] image
  __syncthreads();                    // staging no longer needed
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) {
#pragma unroll
    for (int g = 0; g < 8; ++g) {  // pack row pairs (2g, 2g+1)
      const int d0 = dc * 32 + (2 * g & 3) + 8 * ((2 * g) >> 2) + 4 * hi;
      __bf16 pr[2] = {(__bf16)(oacc[dc][2 * g] * inv_l),
                      (__bf16)(oacc[dc][2 * g + 1] * inv_l)};
      *(uint*)&ow[qcol * OP + d0] = *(uint*)pr;
    }
  }
  // wave-local image; DS ops of one wave are ordered — no barrier needed
  bf16* Op = O + ((long)b * HQ + hq) * D;
  const long sOs = (long)Bb * HQ * D;
  {
    // 64 lanes × 16 B cover 8 q-rows per iteration (D=128 → 2 slices/row)
    constexpr int SLICES = D / 8;  // 16B slices per row
    for (int it = 0; it < 32 * SLICES / 64; ++it) {
      const int flat = it * 64 + lane;
      const int qr = flat / SLICES;
      const int sc = flat % SLICES;
      const int gq = qrow_w + qr;
      if (gq < S) {
        int4 vbits = *(const int4*)&ow[qr * OP + sc * 8];
        *(int4*)(Op + (long)gq * sOs + sc * 8) = vbits;
      }
    }
  }
  if (hi == 0 && my_q < S) {
    float* Lp = LSE + ((long)b * HQ + hq) * S;
    Lp[my_q] = m_i + __logf(fmaxf(l_i, 1e-30f));
  }
}

extern "C" {
void launch_flash_fwd_v3(const void* q, const void* k, const void* v, void* o,
                         void* lse, int B, int HQ, int HKV, int S, int D,
                         bool causal, float scale, int window,
                         const long* qstr, const long* kstr, const long* vstr,
                         hipStream_t stream) {
  dim3 grid((S + 255) / 256, B * HQ);
  dim3 blk(512);
#define CASEV3(DD, CC)                                                        \
 hipLaunchKernelGGL(( flash_fwd_v3_kernel<DD, CC>), dim3(grid), dim3(blk), 0, stream,                       \
      (const bf16*)q, (const bf16*)k, (const bf16*)v, (bf16*)o, (float*)lse, \
      S, B, HQ, HKV, scale, window, qstr[0], qstr[1], qstr[2], kstr[0],       \
      kstr[1], kstr[2], vstr[0], vstr[1], vstr[2])
  if (D == 128) {
    if (causal) CASEV3(128, true); else CASEV3(128, false);
  } else if (D == 64) {
    if (causal) CASEV3(64, true); else CASEV3(64, false);
  }
#undef CASEV3
}
}
