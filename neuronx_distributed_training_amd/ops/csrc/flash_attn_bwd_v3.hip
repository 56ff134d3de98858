// Flash attention backward v3 — T12 swapped-operand structure (DARK:
// dispatched only when NXDT_ATTN_V3=1; see ROADMAP.md §1 and the fwd v3
// kernel). 32×32×16 MFMAs (maps probe-verified), P/dS never touch LDS:
// the wave-half exchange (permlane32_swap) assembles their fragments in
// registers.
//
// dq kernel: wave owns 32 queries (lane = one q column of S^T); loops
//   32-key tiles. S^T = K·Q^T and dP^T = V·dO^T share the C layout, so
//   dS^T is an elementwise register op; dS fragments for dQ = dS^T·...
//   come from the half exchange; dQ = mfma(dS_frags, K^T image).
// dkv kernel: wave owns 32 keys (lane = one key column of S); loops
//   32-query tiles of one (b, hkv, g) slice — grid y enumerates the GQA
//   group like v2, outputs fp32 partial slabs summed by the binding.
//   KNOWN (compile-time): dkv-v3 allocates 256 VGPR + ~200 B/lane scratch
//   (two C accumulator sets + resident K/V frags) — round 2 should split
//   it into separate dV and dK kernels (S recompute is cheaper than the
//   spill) before A/B-ing against v2.
#include "attn_common.h"

typedef float f32x16_t __attribute__((ext_vector_type(16)));

#define MFMA_32x32x16(A, B, C) \
  __builtin_amdgcn_mfma_f32_32x32x16_bf16((A), (B), (C), 0, 0, 0)

// half-exchange: from 16 per-lane f32 values laid out in the 32×32 C map
// (value r ↔ row (r&3)+8(r>>2)+4·hi of this lane's column), build the two
// 8-row A/B fragments per 16-row window (see fwd v3 derivation).
DEVINL void build_halffrags(const float* v, bf16x8_t out[2]) {
  uint pk[8];
#pragma unroll
  for (int g = 0; g < 8; ++g) {
    __bf16 pr[2] = {(__bf16)v[2 * g], (__bf16)v[2 * g + 1]};
    pk[g] = *(uint*)pr;
  }
#pragma unroll
  for (int half = 0; half < 2; ++half) {
    uint a0 = pk[4 * half + 0], a1 = pk[4 * half + 1];
    uint b0 = pk[4 * half + 2], b1 = pk[4 * half + 3];
    auto r0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
    auto r1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
    uint frag[4] = {(uint)r0[0], (uint)r1[0], (uint)r0[1], (uint)r1[1]};
    out[half] = *(bf16x8_t*)frag;
  }
}

// ============================ dQ (v3) ============================
template <int D, bool CAUSAL>
__global__ __launch_bounds__(512) void flash_bwd_dq_v3_kernel(
    const bf16* __restrict__ dO, const bf16* __restrict__ Q,
    const bf16* __restrict__ K, const bf16* __restrict__ V,
    const float* __restrict__ LSE, const float* __restrict__ DELTA,
    bf16* __restrict__ dQ, int S, int Bb, int HQ, int HKV, float scale,
    int window, long sQs, long sQb, long sQh, long sKs, long sKb, long sKh,
    long sVs, long sVb, long sVh, long sDs, long sDb, long sDh) {
  constexpr int BM = 256, BN = 32;
  constexpr int KP = D + 8;
  constexpr int VP = BN + 8;
  __shared__ __bf16 k_lds[BN * KP];
  __shared__ __bf16 v_lds[BN * KP];
  __shared__ __bf16 kt_lds[D * VP];  // K^T (tr_swz) for the dQ B-frags

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int qcol = lane & 31;
  const int qblock = blockIdx.x;
  const int bh = blockIdx.y;
  const int hq = bh % HQ;
  const int b = bh / HQ;
  const int hkv = hq / (HQ / HKV);

  const bf16* Qp = Q + b * sQb + hq * sQh;
  const bf16* dOp = dO + b * sDb + hq * sDh;
  const bf16* Kp = K + b * sKb + hkv * sKh;
  const bf16* Vp = V + b * sVb + hkv * sVh;
  const float* Lp = LSE + ((long)b * HQ + hq) * S;
  const float* Dp = DELTA + ((long)b * HQ + hq) * S;

  const int q0 = qblock * BM;
  const int qrow_w = q0 + wid * 32;
  const int my_q = qrow_w + qcol;

  constexpr int CK = D / 16;
  bf16x8_t qb[CK], dob[CK];  // B-frags of Q^T and dO^T (per own query)
  {
    const long row = (my_q < S) ? my_q : (S - 1);
#pragma unroll
    for (int ck = 0; ck < CK; ++ck) {
      *(int4*)&qb[ck] = *(const int4*)(Qp + row * sQs + ck * 16 + hi * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        qb[ck][j] = (__bf16)((float)qb[ck][j] * scale);  // fold scale
      *(int4*)&dob[ck] = *(const int4*)(dOp + row * sDs + ck * 16 + hi * 8);
    }
  }
  const float lse = (my_q < S) ? Lp[my_q] : 1e30f;
  const float delta = (my_q < S) ? Dp[my_q] : 0.f;

  constexpr int DC = D / 32;
  f32x16_t dqacc[DC];  // C: col = d-in-chunk? no — see below
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) dqacc[dc] = f32x16_t{};
  // NOTE on dqacc layout: dQ = mfma(A = dS frags [32 q × 16 key], B =
  // K^T-image frags [16 key × 32 d]) → C cols = d (lane&31), rows = q.
  // So dqacc holds, per lane, 16 q-rows of ONE d column per 32-d chunk —
  // the per-reg q row is (r&3)+8(r>>2)+4·hi + qrow_w, NOT this lane's
  // my_q. lse/delta for the dS computation are per-q of the S^T layout
  // (my_q), which IS lane-scalar. The epilogue store uses the C map.

  const int kend = CAUSAL ? min(S, q0 + BM) : S;
  const int nkb = (kend + BN - 1) / BN;
  const int jb0 =
      (CAUSAL && window > 0) ? max(0, (q0 - window + 1) / BN) : 0;
  const int wrow_max = qrow_w + 31;

  for (int jb = jb0; jb < nkb; ++jb) {
    const int kbase = jb * BN;
    {  // stage K, V row-major + K^T (tr_swz)
      const int t = threadIdx.x;
      const int row = t / (D / 8);
      const int col8 = (t % (D / 8)) * 8;
      const int gr = kbase + row;
      int4 kv = (gr < S) ? *(const int4*)(Kp + (long)gr * sKs + col8)
                         : int4{0, 0, 0, 0};
      int4 vv = (gr < S) ? *(const int4*)(Vp + (long)gr * sVs + col8)
                         : int4{0, 0, 0, 0};
      *(int4*)&k_lds[row * KP + col8] = kv;
      *(int4*)&v_lds[row * KP + col8] = vv;
      if (t < BN * D / 16) {
        const int prow = (t / (D / 8)) * 2;
        const int pcol8 = (t % (D / 8)) * 8;
        const int g0 = kbase + prow, g1 = g0 + 1;
        int4 k0 = (g0 < S) ? *(const int4*)(Kp + (long)g0 * sKs + pcol8)
                           : int4{0, 0, 0, 0};
        int4 k1 = (g1 < S) ? *(const int4*)(Kp + (long)g1 * sKs + pcol8)
                           : int4{0, 0, 0, 0};
        const __bf16 *e0 = (const __bf16*)&k0, *e1 = (const __bf16*)&k1;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          __bf16 pr[2] = {e0[j], e1[j]};
          const int r = pcol8 + j;
          *(uint*)((char*)kt_lds + tr_swz((uint)(r * VP + prow) * 2, r)) =
              *(uint*)pr;
        }
      }
    }
    __syncthreads();

    if (!CAUSAL || kbase <= wrow_max) {
      // ---- S^T = K·Q^T and dP^T = V·dO^T (same C layout) ----
      f32x16_t st = f32x16_t{};
      f32x16_t dpt = f32x16_t{};
#pragma unroll
      for (int ck = 0; ck < CK; ++ck) {
        bf16x8_t ka, va;
        *(int4*)&ka = *(const int4*)&k_lds[(lane & 31) * KP + ck * 16 + hi * 8];
        *(int4*)&va = *(const int4*)&v_lds[(lane & 31) * KP + ck * 16 + hi * 8];
        st = MFMA_32x32x16(ka, qb[ck], st);
        dpt = MFMA_32x32x16(va, dob[ck], dpt);
      }
      // dS^T = P^T ∘ (dP^T − delta) · scale  (keys per reg as in fwd v3)
      float dsv[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int krow = kbase + (r & 3) + 8 * (r >> 2) + 4 * hi;
        bool dead = (krow >= S) || (my_q >= S) || (CAUSAL && krow > my_q);
        if (CAUSAL && window > 0) dead |= (krow <= my_q - window);
        const float p = dead ? 0.f : __expf(st[r] - lse);
        dsv[r] = p * (dpt[r] - delta) * scale;
      }
      // assemble dS A-frags (rows = q) via the half exchange:
      // build_halffrags gives, per 16-key window, the two key-chunks of
      // this lane's q column redistributed so lane holds 8 consecutive
      // keys — exactly the A[row=q][k=key] fragment (A row = lane&31 = q).
      bf16x8_t dsf[2];
      build_halffrags(dsv, dsf);
      // ---- dQ += dS · K : B = K^T image frags ----
#pragma unroll
      for (int dc = 0; dc < DC; ++dc) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          bf16x8_t kb;  // B[k=key=16kk+hi*8+j][j=d=dc*32+(lane&31)]
          const int drow = dc * 32 + (lane & 31);
          const uint byte =
              tr_swz((uint)(drow * VP + kk * 16 + hi * 8) * 2, drow);
          *(int4*)&kb = *(const int4*)((const char*)kt_lds + byte);
          dqacc[dc] = MFMA_32x32x16(dsf[kk], kb, dqacc[dc]);
        }
      }
    }
    __syncthreads();
  }

  // ---- store dQ: C map (col = d = lane&31 within chunk, rows = q) ----
  bf16* dQp = dQ + ((long)b * HQ + hq) * D;
  const long sOs = (long)Bb * HQ * D;
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = qrow_w + (r & 3) + 8 * (r >> 2) + 4 * hi;
      if (qrow < S)
        dQp[(long)qrow * sOs + dc * 32 + (lane & 31)] = f2bf(dqacc[dc][r]);
    }
  }
}

// ============================ dK/dV (v3) ============================
// Split into two single-output instantiations (WHICH: 0 = dV, 1 = dK) —
// a fused dkv at this tiling allocates 256 VGPR + ~200 B/lane scratch;
// recomputing S in a second launch is cheaper than the spill.
template <int D, bool CAUSAL, int WHICH>
__global__ __launch_bounds__(512) void flash_bwd_dkv_v3_kernel(
    const bf16* __restrict__ dO, const bf16* __restrict__ Q,
    const bf16* __restrict__ K, const bf16* __restrict__ V,
    const float* __restrict__ LSE, const float* __restrict__ DELTA,
    float* __restrict__ OUT,  // [group][s][b][hkv][d] fp32 partials
    int S, int Bb, int HQ, int HKV, float scale, int window, long sQs,
    long sQb, long sQh, long sKs, long sKb, long sKh, long sVs, long sVb,
    long sVh, long sDs, long sDb, long sDh) {
  constexpr int BN = 256, BM = 32;  // keys per block, q tile
  constexpr int KP = D + 8;
  constexpr int VP = BM + 8;
  __shared__ __bf16 q_lds[BM * KP];
  __shared__ __bf16 aux_lds[BM * KP];  // dO rows (WHICH==1 only)
  __shared__ __bf16 t_lds[D * VP];     // dO^T (dV) / Q^T (dK), tr_swz

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int kcol = lane & 31;
  const int kblock = blockIdx.x;
  const int group = HQ / HKV;
  const int bhg = blockIdx.y;  // (b * HKV + hkv) * group + g
  const int g = bhg % group;
  const int hkv = (bhg / group) % HKV;
  const int b = bhg / (group * HKV);
  const int hq = hkv * group + g;

  const bf16* Kp = K + b * sKb + hkv * sKh;
  const bf16* Vp = V + b * sVb + hkv * sVh;
  const bf16* Qp = Q + b * sQb + hq * sQh;
  const bf16* dOp = dO + b * sDb + hq * sDh;
  const float* Lp = LSE + ((long)b * HQ + hq) * S;
  const float* Dp = DELTA + ((long)b * HQ + hq) * S;

  const int kbase = kblock * BN;
  const int krow_w = kbase + wid * 32;
  const int my_k = krow_w + kcol;

  constexpr int CK = D / 16;
  bf16x8_t kb[CK];           // K^T B-frags (own key row, scale folded)
  bf16x8_t vb[WHICH ? CK : 1];  // V^T B-frags only needed for dK (dP)
  {
    const long row = (my_k < S) ? my_k : (S - 1);
#pragma unroll
    for (int ck = 0; ck < CK; ++ck) {
      *(int4*)&kb[ck] = *(const int4*)(Kp + row * sKs + ck * 16 + hi * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        kb[ck][j] = (__bf16)((float)kb[ck][j] * scale);
      if constexpr (WHICH == 1)
        *(int4*)&vb[ck] = *(const int4*)(Vp + row * sVs + ck * 16 + hi * 8);
    }
  }

  constexpr int DC = D / 32;
  f32x16_t acc[DC];  // C: col = d, rows = key
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) acc[dc] = f32x16_t{};

  const int ib0 = CAUSAL ? (kbase / BM) : 0;
  int nqb = (S + BM - 1) / BM;
  if (CAUSAL && window > 0)
    nqb = min(nqb, (kbase + BN - 1 + window + BM - 1) / BM);
  const int wkey_min = krow_w;

  for (int ib = ib0; ib < nqb; ++ib) {
    const int qbase = ib * BM;
    {  // stage: q rows always; dO rows (dK); dO^T or Q^T image
      const int t = threadIdx.x;
      const int row = t / (D / 8);
      const int col8 = (t % (D / 8)) * 8;
      const int gr = qbase + row;
      int4 qv = (gr < S) ? *(const int4*)(Qp + (long)gr * sQs + col8)
                         : int4{0, 0, 0, 0};
      *(int4*)&q_lds[row * KP + col8] = qv;
      if constexpr (WHICH == 1) {
        int4 dv = (gr < S) ? *(const int4*)(dOp + (long)gr * sDs + col8)
                           : int4{0, 0, 0, 0};
        *(int4*)&aux_lds[row * KP + col8] = dv;
      }
      if (t < BM * D / 16) {
        const int prow = (t / (D / 8)) * 2;
        const int pcol8 = (t % (D / 8)) * 8;
        const int g0 = qbase + prow, g1 = g0 + 1;
        const bf16* src = WHICH ? Qp : dOp;
        const long stride = WHICH ? sQs : sDs;
        int4 a0 = (g0 < S) ? *(const int4*)(src + (long)g0 * stride + pcol8)
                           : int4{0, 0, 0, 0};
        int4 a1 = (g1 < S) ? *(const int4*)(src + (long)g1 * stride + pcol8)
                           : int4{0, 0, 0, 0};
        const __bf16 *e0 = (const __bf16*)&a0, *e1 = (const __bf16*)&a1;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          __bf16 pr[2] = {e0[j], e1[j]};
          const int r = pcol8 + j;
          *(uint*)((char*)t_lds + tr_swz((uint)(r * VP + prow) * 2, r)) =
              *(uint*)pr;
        }
      }
    }
    __syncthreads();

    if (!CAUSAL || wkey_min <= qbase + BM - 1) {
      // ---- S = Q·K^T (C: col = key, rows = q); dP = dO·V^T (dK) ----
      f32x16_t st = f32x16_t{};
      f32x16_t dpt = f32x16_t{};
#pragma unroll
      for (int ck = 0; ck < CK; ++ck) {
        bf16x8_t qa;
        *(int4*)&qa = *(const int4*)&q_lds[(lane & 31) * KP + ck * 16 + hi * 8];
        st = MFMA_32x32x16(qa, kb[ck], st);
        if constexpr (WHICH == 1) {
          bf16x8_t da;
          *(int4*)&da =
              *(const int4*)&aux_lds[(lane & 31) * KP + ck * 16 + hi * 8];
          dpt = MFMA_32x32x16(da, vb[ck], dpt);
        }
      }
      float val[16];  // P (dV) or dS (dK), per q row of the C map
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = qbase + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const float lse = (qrow < S) ? Lp[qrow] : 1e30f;
        bool dead = (qrow >= S) || (my_k >= S) || (CAUSAL && my_k > qrow);
        if (CAUSAL && window > 0) dead |= (my_k <= qrow - window);
        const float p = dead ? 0.f : __expf(st[r] - lse);
        if constexpr (WHICH == 1) {
          const float delta = (qrow < S) ? Dp[qrow] : 0.f;
          val[r] = p * (dpt[r] - delta) * scale;
        } else {
          val[r] = p;
        }
      }
      // A-frags (rows = key = lane&31) via the wave-half exchange
      bf16x8_t af[2];
      build_halffrags(val, af);
      // ---- acc += P^T·dO (dV) or dS^T·Q (dK) ----
#pragma unroll
      for (int dc = 0; dc < DC; ++dc) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          const int drow = dc * 32 + (lane & 31);
          const uint byte =
              tr_swz((uint)(drow * VP + kk * 16 + hi * 8) * 2, drow);
          bf16x8_t bf;
          *(int4*)&bf = *(const int4*)((const char*)t_lds + byte);
          acc[dc] = MFMA_32x32x16(af[kk], bf, acc[dc]);
        }
      }
    }
    __syncthreads();
  }

  // ---- store fp32 partials [group][s][b][hkv][d]; C rows = key ----
  const long sOs = (long)Bb * HKV * D;
  float* Op = OUT + (long)g * S * sOs + ((long)b * HKV + hkv) * D;
#pragma unroll
  for (int dc = 0; dc < DC; ++dc) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krow = krow_w + (r & 3) + 8 * (r >> 2) + 4 * hi;
      if (krow < S)
        Op[(long)krow * sOs + dc * 32 + (lane & 31)] = acc[dc][r];
    }
  }
}

extern "C" {
void launch_flash_bwd_v3(const void* dout, const void* q, const void* k,
                         const void* v, const void* lse, const void* delta,
                         void* dq, void* dk, void* dv, int B, int HQ,
                         int HKV, int S, int D, bool causal, float scale,
                         int window, const long* qstr, const long* kstr,
                         const long* vstr, const long* dostr,
                         hipStream_t stream) {
  dim3 blk(512);
  dim3 gkv((S + 255) / 256, B * HQ);
  dim3 gq((S + 255) / 256, B * HQ);
#define CASEB3(DD, CC)                                                        \
  do {                                                                        \
    flash_bwd_dkv_v3_kernel<DD, CC, 0><<<gkv, blk, 0, stream>>>(              \
        (const bf16*)dout, (const bf16*)q, (const bf16*)k, (const bf16*)v,    \
        (const float*)lse, (const float*)delta, (float*)dv, S, B, HQ, HKV,    \
        scale, window, qstr[0], qstr[1], qstr[2], kstr[0], kstr[1], kstr[2],  \
        vstr[0], vstr[1], vstr[2], dostr[0], dostr[1], dostr[2]);             \
    flash_bwd_dkv_v3_kernel<DD, CC, 1><<<gkv, blk, 0, stream>>>(              \
        (const bf16*)dout, (const bf16*)q, (const bf16*)k, (const bf16*)v,    \
        (const float*)lse, (const float*)delta, (float*)dk, S, B, HQ, HKV,    \
        scale, window, qstr[0], qstr[1], qstr[2], kstr[0], kstr[1], kstr[2],  \
        vstr[0], vstr[1], vstr[2], dostr[0], dostr[1], dostr[2]);             \
    flash_bwd_dq_v3_kernel<DD, CC><<<gq, blk, 0, stream>>>(                   \
        (const bf16*)dout, (const bf16*)q, (const bf16*)k, (const bf16*)v,    \
        (const float*)lse, (const float*)delta, (bf16*)dq, S, B, HQ, HKV,     \
        scale, window, qstr[0], qstr[1], qstr[2], kstr[0], kstr[1], kstr[2],  \
        vstr[0], vstr[1], vstr[2], dostr[0], dostr[1], dostr[2]);             \
  } while (0)
  if (D == 128) {
    if (causal) CASEB3(128, true); else CASEB3(128, false);
  } else if (D == 64) {
    if (causal) CASEB3(64, true); else CASEB3(64, false);
  }
#undef CASEB3
}
}
