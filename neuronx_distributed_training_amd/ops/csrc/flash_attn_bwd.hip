// Causal flash attention BACKWARD for MI355X (gfx950).
// FlashAttention-2 style recompute, two kernels, no atomics:
//   dkv: grid over key blocks  — recompute P^T, accumulate dV, dK
//   dq : grid over query blocks — recompute P,  accumulate dQ
// GQA: dkv sums over the q-head group of each kv head in-kernel.
// delta = rowsum(dO ∘ O) is computed by the caller (eager, tiny).
#include "attn_common.h"

// ============================ dK / dV ============================
// Block: 4 waves, BN=64 keys (16/wave). Loops q heads in the GQA group ×
// q blocks ≥ key block (causal).
template <int D, bool CAUSAL>
__global__ __launch_bounds__(256) void flash_bwd_dkv_kernel(
    const bf16* __restrict__ dO,  // [B, HQ, S, D]
    const bf16* __restrict__ Q,   // [B, HQ, S, D]
    const bf16* __restrict__ K,   // [B, HKV, S, D]
    const bf16* __restrict__ V,   // [B, HKV, S, D]
    const float* __restrict__ LSE,    // [B, HQ, S]
    const float* __restrict__ DELTA,  // [B, HQ, S]
    bf16* __restrict__ dK,  // [B, HKV, S, D]
    bf16* __restrict__ dV,  // [B, HKV, S, D]
    int S, int HQ, int HKV, float scale) {
  constexpr int BM = 64, BN = 64;
  constexpr int KP = D + 8;
  constexpr int VP = BM + 8;
  __shared__ __bf16 q_lds[BM * KP];    // Q rows      (B for S^T)
  __shared__ __bf16 do_lds[BM * KP];   // dO rows     (B for dP^T)
  __shared__ __bf16 dot_lds[D * VP];   // dO^T        (B for dV)
  __shared__ __bf16 qt_lds[D * VP];    // Q^T         (B for dK)
  __shared__ __bf16 pT_lds[4 * 16 * VP];   // per-wave P^T  (A for dV)
  __shared__ __bf16 dsT_lds[4 * 16 * VP];  // per-wave dS^T (A for dK)

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int kblock = blockIdx.x;
  const int bh = blockIdx.y;  // b * HKV + hkv
  const int hkv = bh % HKV;
  const int b = bh / HKV;
  const int group = HQ / HKV;

  const long kvoff = ((long)b * HKV + hkv) * S * D;
  const bf16* Kp = K + kvoff;
  const bf16* Vp = V + kvoff;
  const int kbase = kblock * BN;
  const int krow_w = kbase + wid * 16;

  // K/V fragments for this wave's 16 keys (A-layout rows)
  constexpr int DK = D / 32;
  bf16x8_t kfrag[DK], vfrag[DK];
  {
    const int r = krow_w + (lane & 15);
    const int row = (r < S) ? r : (S - 1);
#pragma unroll
    for (int kk = 0; kk < DK; ++kk) {
      *(int4*)&kfrag[kk] =
          *(const int4*)(Kp + (long)row * D + kk * 32 + (lane >> 4) * 8);
      *(int4*)&vfrag[kk] =
          *(const int4*)(Vp + (long)row * D + kk * 32 + (lane >> 4) * 8);
    }
  }

  constexpr int DN = D / 16;
  f32x4_t dvacc[DN], dkacc[DN];
#pragma unroll
  for (int nj = 0; nj < DN; ++nj) {
    dvacc[nj] = f32x4_t{0.f, 0.f, 0.f, 0.f};
    dkacc[nj] = f32x4_t{0.f, 0.f, 0.f, 0.f};
  }

  for (int g = 0; g < group; ++g) {
    const int hq = hkv * group + g;
    const long qoff = ((long)b * HQ + hq) * S * D;
    const bf16* Qp = Q + qoff;
    const bf16* dOp = dO + qoff;
    const float* Lp = LSE + ((long)b * HQ + hq) * S;
    const float* Dp = DELTA + ((long)b * HQ + hq) * S;

    const int ib0 = CAUSAL ? kblock : 0;  // BM == BN
    const int nqb = (S + BM - 1) / BM;
    for (int ib = ib0; ib < nqb; ++ib) {
      const int qbase = ib * BM;
      // ---- stage Q, dO (row-major + transposed) ----
      {
        constexpr int VECS = BM * D / 8;
        for (int t = threadIdx.x; t < VECS; t += 256) {
          const int row = t / (D / 8);
          const int col8 = (t % (D / 8)) * 8;
          const int gr = qbase + row;
          int4 qv, dv;
          if (gr < S) {
            qv = *(const int4*)(Qp + (long)gr * D + col8);
            dv = *(const int4*)(dOp + (long)gr * D + col8);
          } else {
            qv = int4{0, 0, 0, 0};
            dv = int4{0, 0, 0, 0};
          }
          *(int4*)&q_lds[row * KP + col8] = qv;
          *(int4*)&do_lds[row * KP + col8] = dv;
          const __bf16* qe = (const __bf16*)&qv;
          const __bf16* de = (const __bf16*)&dv;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            qt_lds[(col8 + j) * VP + row] = qe[j];
            dot_lds[(col8 + j) * VP + row] = de[j];
          }
        }
      }
      __syncthreads();

      // ---- per q-chunk: S^T, P^T, dP^T, dS^T ----
      __bf16* pw = &pT_lds[wid * 16 * VP];
      __bf16* dw = &dsT_lds[wid * 16 * VP];
#pragma unroll
      for (int nq = 0; nq < 4; ++nq) {
        f32x4_t st = f32x4_t{0.f, 0.f, 0.f, 0.f};
        f32x4_t dpt = f32x4_t{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kk = 0; kk < DK; ++kk) {
          bf16x8_t qb =
              load_frag_b_rowmajorT(&q_lds[nq * 16 * KP], KP, kk * 32, lane);
          st = MFMA_16x16x32(kfrag[kk], qb, st);
          bf16x8_t db =
              load_frag_b_rowmajorT(&do_lds[nq * 16 * KP], KP, kk * 32, lane);
          dpt = MFMA_16x16x32(vfrag[kk], db, dpt);
        }
        const int qcol = qbase + nq * 16 + (lane & 15);
        const float lse = (qcol < S) ? Lp[qcol] : 1e30f;
        const float delta = (qcol < S) ? Dp[qcol] : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int krow = krow_w + (lane >> 4) * 4 + r;
          float s = st[r] * scale;
          const bool dead =
              (qcol >= S) || (krow >= S) || (CAUSAL && qcol < krow);
          const float p = dead ? 0.f : __expf(s - lse);
          const float ds = p * (dpt[r] - delta) * scale;
          pw[((lane >> 4) * 4 + r) * VP + nq * 16 + (lane & 15)] = (__bf16)p;
          dw[((lane >> 4) * 4 + r) * VP + nq * 16 + (lane & 15)] = (__bf16)ds;
        }
      }
      // own-wave LDS writes are ordered before own-wave reads; tiles
      // (q_lds etc.) are stable until the barrier at loop end.

      // ---- dV += P^T dO ; dK += dS^T Q ----
#pragma unroll
      for (int nj = 0; nj < DN; ++nj) {
#pragma unroll
        for (int kk = 0; kk < BM / 32; ++kk) {
          bf16x8_t pa = load_frag_a(pw, VP, kk * 32, lane);
          bf16x8_t dob =
              load_frag_b_rowmajorT(&dot_lds[nj * 16 * VP], VP, kk * 32, lane);
          dvacc[nj] = MFMA_16x16x32(pa, dob, dvacc[nj]);
          bf16x8_t da = load_frag_a(dw, VP, kk * 32, lane);
          bf16x8_t qb2 =
              load_frag_b_rowmajorT(&qt_lds[nj * 16 * VP], VP, kk * 32, lane);
          dkacc[nj] = MFMA_16x16x32(da, qb2, dkacc[nj]);
        }
      }
      __syncthreads();
    }
  }

  // ---- store dK, dV ----
  bf16* dKp = dK + kvoff;
  bf16* dVp = dV + kvoff;
#pragma unroll
  for (int nj = 0; nj < DN; ++nj) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int krow = krow_w + (lane >> 4) * 4 + r;
      if (krow < S) {
        dKp[(long)krow * D + nj * 16 + (lane & 15)] = f2bf(dkacc[nj][r]);
        dVp[(long)krow * D + nj * 16 + (lane & 15)] = f2bf(dvacc[nj][r]);
      }
    }
  }
}

// ============================ dQ ============================
template <int D, bool CAUSAL>
__global__ __launch_bounds__(256) void flash_bwd_dq_kernel(
    const bf16* __restrict__ dO, const bf16* __restrict__ Q,
    const bf16* __restrict__ K, const bf16* __restrict__ V,
    const float* __restrict__ LSE, const float* __restrict__ DELTA,
    bf16* __restrict__ dQ, int S, int HQ, int HKV, float scale) {
  constexpr int BM = 64, BN = 64;
  constexpr int KP = D + 8;
  constexpr int VP = BN + 8;
  __shared__ __bf16 k_lds[BN * KP];       // K rows   (B for S)
  __shared__ __bf16 v_lds[BN * KP];       // V rows   (B for dP)
  __shared__ __bf16 kt_lds[D * VP];       // K^T      (B for dQ)
  __shared__ __bf16 ds_lds[4 * 16 * VP];  // per-wave dS (A for dQ)

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
  const bf16* dOp = dO + qoff;
  const bf16* Kp = K + kvoff;
  const bf16* Vp = V + kvoff;
  const float* Lp = LSE + ((long)b * HQ + hq) * S;
  const float* Dp = DELTA + ((long)b * HQ + hq) * S;

  const int q0 = qblock * BM;
  const int qrow_w = q0 + wid * 16;

  constexpr int DK = D / 32;
  bf16x8_t qfrag[DK], dofrag[DK];
  {
    const int r = qrow_w + (lane & 15);
    const int row = (r < S) ? r : (S - 1);
#pragma unroll
    for (int kk = 0; kk < DK; ++kk) {
      *(int4*)&qfrag[kk] =
          *(const int4*)(Qp + (long)row * D + kk * 32 + (lane >> 4) * 8);
      *(int4*)&dofrag[kk] =
          *(const int4*)(dOp + (long)row * D + kk * 32 + (lane >> 4) * 8);
    }
  }
  float lse[4], delta[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = qrow_w + (lane >> 4) * 4 + r;
    lse[r] = (qrow < S) ? Lp[qrow] : 1e30f;
    delta[r] = (qrow < S) ? Dp[qrow] : 0.f;
  }

  constexpr int DN = D / 16;
  f32x4_t dqacc[DN];
#pragma unroll
  for (int nj = 0; nj < DN; ++nj) dqacc[nj] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const int kend = CAUSAL ? min(S, q0 + BM) : S;
  const int nkb = (kend + BN - 1) / BN;
  for (int jb = 0; jb < nkb; ++jb) {
    const int kbase = jb * BN;
    {
      constexpr int VECS = BN * D / 8;
      for (int t = threadIdx.x; t < VECS; t += 256) {
        const int row = t / (D / 8);
        const int col8 = (t % (D / 8)) * 8;
        const int gr = kbase + row;
        int4 kv, vv;
        if (gr < S) {
          kv = *(const int4*)(Kp + (long)gr * D + col8);
          vv = *(const int4*)(Vp + (long)gr * D + col8);
        } else {
          kv = int4{0, 0, 0, 0};
          vv = int4{0, 0, 0, 0};
        }
        *(int4*)&k_lds[row * KP + col8] = kv;
        *(int4*)&v_lds[row * KP + col8] = vv;
        const __bf16* ke = (const __bf16*)&kv;
#pragma unroll
        for (int j = 0; j < 8; ++j) kt_lds[(col8 + j) * VP + row] = ke[j];
      }
    }
    __syncthreads();

    __bf16* dsw = &ds_lds[wid * 16 * VP];
#pragma unroll
    for (int nk = 0; nk < 4; ++nk) {
      f32x4_t st = f32x4_t{0.f, 0.f, 0.f, 0.f};
      f32x4_t dpt = f32x4_t{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < DK; ++kk) {
        bf16x8_t kb =
            load_frag_b_rowmajorT(&k_lds[nk * 16 * KP], KP, kk * 32, lane);
        st = MFMA_16x16x32(qfrag[kk], kb, st);
        bf16x8_t vb =
            load_frag_b_rowmajorT(&v_lds[nk * 16 * KP], KP, kk * 32, lane);
        dpt = MFMA_16x16x32(dofrag[kk], vb, dpt);
      }
      const int kcol = kbase + nk * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = qrow_w + (lane >> 4) * 4 + r;
        const bool dead = (kcol >= S) || (CAUSAL && kcol > qrow);
        const float p = dead ? 0.f : __expf(st[r] * scale - lse[r]);
        const float ds = p * (dpt[r] - delta[r]) * scale;
        dsw[((lane >> 4) * 4 + r) * VP + nk * 16 + (lane & 15)] = (__bf16)ds;
      }
    }

    // dQ += dS K  (A = dS 16×BN, B = K BN×D via K^T image)
#pragma unroll
    for (int nj = 0; nj < DN; ++nj) {
#pragma unroll
      for (int kk = 0; kk < BN / 32; ++kk) {
        bf16x8_t da = load_frag_a(dsw, VP, kk * 32, lane);
        bf16x8_t kb2 =
            load_frag_b_rowmajorT(&kt_lds[nj * 16 * VP], VP, kk * 32, lane);
        dqacc[nj] = MFMA_16x16x32(da, kb2, dqacc[nj]);
      }
    }
    __syncthreads();
  }

  bf16* dQp = dQ + qoff;
#pragma unroll
  for (int nj = 0; nj < DN; ++nj) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = qrow_w + (lane >> 4) * 4 + r;
      if (qrow < S)
        dQp[(long)qrow * D + nj * 16 + (lane & 15)] = f2bf(dqacc[nj][r]);
    }
  }
}

extern "C" {
void launch_flash_bwd(const void* dout, const void* q, const void* k,
                      const void* v, const void* lse, const void* delta,
                      void* dq, void* dk, void* dv, int B, int HQ, int HKV,
                      int S, int D, bool causal, float scale,
                      hipStream_t stream) {
  dim3 blk(256);
  dim3 gkv((S + 63) / 64, B * HKV);
  dim3 gq((S + 63) / 64, B * HQ);
#define CASE(DD, CC)                                                          \
  do {                                                                        \
    flash_bwd_dkv_kernel<DD, CC><<<gkv, blk, 0, stream>>>(                    \
        (const bf16*)dout, (const bf16*)q, (const bf16*)k, (const bf16*)v,    \
        (const float*)lse, (const float*)delta, (bf16*)dk, (bf16*)dv, S, HQ,  \
        HKV, scale);                                                          \
    flash_bwd_dq_kernel<DD, CC><<<gq, blk, 0, stream>>>(                      \
        (const bf16*)dout, (const bf16*)q, (const bf16*)k, (const bf16*)v,    \
        (const float*)lse, (const float*)delta, (bf16*)dq, S, HQ, HKV,        \
        scale);                                                               \
  } while (0)
  if (D == 128) {
    if (causal) CASE(128, true); else CASE(128, false);
  } else if (D == 64) {
    if (causal) CASE(64, true); else CASE(64, false);
  }
#undef CASE
}
}
