// DOUBLE-BUFFERED variant of flash_fwd_kernel (dark; A/B via the
// flash_attn_fwd_dbuf binding): two K/V LDS buffer sets -> ONE barrier
// per tile and the staging writes overlap the previous tile's MFMAs.
// LDS 108.5 KB -> 1 block/CU (8 waves) vs the default's 2 blocks; which
// side wins is an empirical question (the dkv kernel lives at 1 block).
#include "attn_common.h"

// NSB = 16-row sub-blocks per wave: 2 -> BM 256 (the TP<=4 shape), 1 ->
// BM 128 for small B*HQ launches (TP=8 has HQ_local=4: BM=256 gives 256
// blocks = 1/CU on the bench shape — half the chip idles).
template <int D, bool CAUSAL, int NSB>
__global__ __launch_bounds__(512) void flash_fwd_dbuf_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ K,
    const bf16* __restrict__ V, bf16* __restrict__ O,
    float* __restrict__ LSE,  // [B, HQ, SQ] f32
    int SQ, int SKV, int Bb, int HQ, int HKV, float scale,
    int window,                     // sliding window (<=0: disabled)
    long sQs, long sQb, long sQh,   // Q element strides (seq, batch, head)
    long sKs, long sKb, long sKh,   // K strides
    long sVs, long sVb, long sVh) { // V strides
  constexpr int BM = NSB * 128, BN = 64;
  constexpr int KP = D + 8;
  constexpr int VP = BN + 8;
  __shared__ __bf16 k_lds[2][BN * KP];
  __shared__ __bf16 vt_lds[2][D * VP];
  __shared__ __bf16 p_lds[8 * NSB * 16 * VP];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  int qblock, bh;  // b * HQ + hq
  xcd_remap(qblock, bh);
  const int hq = bh % HQ;
  const int b = bh / HQ;
  const int hkv = hq / (HQ / HKV);
  // S_q != S_kv: causal is BOTTOM-RIGHT aligned (query i sees keys
  // j <= i + SKV - SQ) — the KV-cache decode / ring half-block convention
  const int coff = SKV - SQ;

  const bf16* Qp = Q + b * sQb + hq * sQh;
  const bf16* Kp = K + b * sKb + hkv * sKh;
  const bf16* Vp = V + b * sVb + hkv * sVh;

  const int q0 = qblock * BM;
  const int qrow_w = q0 + wid * NSB * 16;  // first row of this wave

  constexpr int DK = D / 32;
  bf16x8_t qfrag[NSB][DK];
#pragma unroll
  for (int sb = 0; sb < NSB; ++sb) {
    const int r = qrow_w + sb * 16 + (lane & 15);
    const long row = (r < SQ) ? r : (SQ - 1);
#pragma unroll
    for (int kk = 0; kk < DK; ++kk) {
      const bf16* p = Qp + row * sQs + kk * 32 + (lane >> 4) * 8;
      *(int4*)&qfrag[sb][kk] = *(const int4*)p;
      // fold the softmax scale into Q once (saves a VALU mul per score
      // per tile; PMC: these kernels are VALU-bound at ~8 VALU/MFMA)
#pragma unroll
      for (int j = 0; j < 8; ++j)
        qfrag[sb][kk][j] = (__bf16)((float)qfrag[sb][kk][j] * scale);
    }
  }

  float m_i[NSB][4], l_i[NSB][4];
  float alpha_s[NSB][4];
  // all-ones B fragment: one MFMA per 32-key chunk computes the P row-sums
  // into every lane's accumulator (replaces 16 adds + 16 shuffles per
  // sub-block of VALU reduction)
  bf16x8_t ones_frag;
#pragma unroll
  for (int j = 0; j < 8; ++j) ones_frag[j] = (__bf16)1.0f;
  constexpr int DN = D / 16;
  f32x4_t oacc[NSB][DN];
#pragma unroll
  for (int sb = 0; sb < NSB; ++sb) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_i[sb][r] = -1e30f;
      l_i[sb][r] = 0.f;
    }
#pragma unroll
    for (int nj = 0; nj < DN; ++nj) oacc[sb][nj] = f32x4_t{0.f, 0.f, 0.f, 0.f};
  }

  const int kend = CAUSAL ? min(SKV, q0 + BM + coff) : SKV;
  const int nkb = (kend + BN - 1) / BN;
  const int jb0 =
      (CAUSAL && window > 0) ? max(0, (q0 + coff - window + 1) / BN) : 0;
  const int wrow_max = qrow_w + NSB * 16 - 1;

  // T5 static form: the younger dispatch half gets priority so it is not
  // starved of VALU issue at segment starts (guide §5.5 T5).
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  // T14 async-stage split: issue tile t+1's global loads into registers
  // BEFORE computing tile t (HBM latency hides under the MFMAs) and write
  // them to LDS only after the end-of-tile barrier (guide §5.5 T14/G15).
  // Per-thread staging registers: K = 2 int4 rows-slices, V = 1 row-pair.
  constexpr int KV_PER_THR = BN * D / 8 / 512;  // int4 K vectors / thread
  int4 kreg[KV_PER_THR];
  int4 vreg0, vreg1;

  auto issue_loads = [&](int kbase) {
#pragma unroll
    for (int u = 0; u < KV_PER_THR; ++u) {
      const int t = threadIdx.x + u * 512;
      const int row = t / (D / 8);
      const int col8 = (t % (D / 8)) * 8;
      const int gr = kbase + row;
      kreg[u] = (gr < SKV) ? *(const int4*)(Kp + (long)gr * sKs + col8)
                           : int4{0, 0, 0, 0};
    }
    {
      const int t = threadIdx.x;  // BN*D/16 == 512 row-pair slices
      const int row = (t / (D / 8)) * 2;
      const int col8 = (t % (D / 8)) * 8;
      const int g0 = kbase + row, g1 = g0 + 1;
      vreg0 = (g0 < SKV) ? *(const int4*)(Vp + (long)g0 * sVs + col8)
                         : int4{0, 0, 0, 0};
      vreg1 = (g1 < SKV) ? *(const int4*)(Vp + (long)g1 * sVs + col8)
                         : int4{0, 0, 0, 0};
    }
  };

  auto write_lds = [&](int buf) {
#pragma unroll
    for (int u = 0; u < KV_PER_THR; ++u) {
      const int t = threadIdx.x + u * 512;
      const int row = t / (D / 8);
      const int col8 = (t % (D / 8)) * 8;
      *(int4*)&k_lds[buf][row * KP + col8] = kreg[u];
    }
    {
      const int t = threadIdx.x;
      const int row = (t / (D / 8)) * 2;
      const int col8 = (t % (D / 8)) * 8;
      const __bf16* e0 = (const __bf16*)&vreg0;
      const __bf16* e1 = (const __bf16*)&vreg1;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        __bf16 pair[2] = {e0[j], e1[j]};
        const int r = col8 + j;
        *(uint*)((char*)vt_lds[buf] + tr_swz((uint)(r * VP + row) * 2, r)) =
            *(uint*)pair;
      }
    }
  };

  issue_loads(jb0 * BN);
  write_lds(jb0 & 1);
  __syncthreads();

  for (int jb = jb0; jb < nkb; ++jb) {
    const int kbase = jb * BN;
    const int cur = jb & 1;
    if (jb + 1 < nkb) issue_loads((jb + 1) * BN);

    if (!CAUSAL || kbase <= wrow_max + coff) {
      // ---- S = Q K^T for both sub-blocks (B-frags loaded once) ----
      f32x4_t sacc[NSB][4];
#pragma unroll
      for (int nk = 0; nk < 4; ++nk) {
#pragma unroll
        for (int sb = 0; sb < NSB; ++sb)
          sacc[sb][nk] = f32x4_t{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kk = 0; kk < DK; ++kk) {
          bf16x8_t bfrag =
              load_frag_b_rowmajorT(&k_lds[cur][nk * 16 * KP], KP, kk * 32, lane);
#pragma unroll
          for (int sb = 0; sb < NSB; ++sb)
            sacc[sb][nk] = MFMA_16x16x32(qfrag[sb][kk], bfrag, sacc[sb][nk]);
        }
      }
      // ---- mask + online softmax + P→LDS, per sub-block ----
      __bf16* pw = &p_lds[wid * NSB * 16 * VP];
#pragma unroll
      for (int sb = 0; sb < NSB; ++sb) {
        // interior tiles (every key visible to every row) skip the mask
        const bool full_tile =
            (kbase + BN <= SKV) &&
            (!CAUSAL || (kbase + BN - 1 <= qrow_w + sb * 16 + coff)) &&
            (window <= 0 || kbase >= qrow_w + sb * 16 + coff + 15 - window + 1);
        float tile_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
        float sv[4][4];
        if (full_tile) {
#pragma unroll
          for (int nk = 0; nk < 4; ++nk) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              const float s = sacc[sb][nk][r];  // scale folded into Q
              sv[nk][r] = s;
              tile_max[r] = fmaxf(tile_max[r], s);
            }
          }
        } else {
#pragma unroll
          for (int nk = 0; nk < 4; ++nk) {
            const int kcol = kbase + nk * 16 + (lane & 15);
#pragma unroll
            for (int r = 0; r < 4; ++r) {
              const int qrow = qrow_w + sb * 16 + (lane >> 4) * 4 + r;
              float s = sacc[sb][nk][r];  // scale folded into Q
              bool dead = (kcol >= SKV) || (CAUSAL && kcol > qrow + coff);
              if (CAUSAL && window > 0) dead |= (kcol <= qrow + coff - window);
              s = dead ? -1e30f : s;
              sv[nk][r] = s;
              tile_max[r] = fmaxf(tile_max[r], s);
            }
          }
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
#pragma unroll
          for (int off = 1; off < 16; off <<= 1)
            tile_max[r] = fmaxf(tile_max[r], __shfl_xor(tile_max[r], off, 64));
        }
        // rescale is EXACTLY the identity when no row's max grew — skip
        // the alpha exps and the O/l rescale entirely then (wave-uniform
        // vote; most interior tiles after the first few don't move m).
        bool grew = false;
#pragma unroll
        for (int r = 0; r < 4; ++r) grew |= tile_max[r] > m_i[sb][r];
        if (__builtin_amdgcn_ballot_w64(grew) != 0ull) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float mn = fmaxf(m_i[sb][r], tile_max[r]);
            alpha_s[sb][r] = __expf(m_i[sb][r] - mn);
            m_i[sb][r] = mn;
          }
#pragma unroll
          for (int nj = 0; nj < DN; ++nj) {
#pragma unroll
            for (int r = 0; r < 4; ++r) oacc[sb][nj][r] *= alpha_s[sb][r];
          }
        } else {
#pragma unroll
          for (int r = 0; r < 4; ++r) alpha_s[sb][r] = 1.0f;
        }
#pragma unroll
        for (int nk = 0; nk < 4; ++nk) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const float p = __expf(sv[nk][r] - m_i[sb][r]);
            pw[(sb * 16 + (lane >> 4) * 4 + r) * VP + nk * 16 + (lane & 15)] =
                (__bf16)p;
          }
        }
      }
      // ---- O += P V (V^T B-frags loaded once per sub-block set) ----
      f32x4_t racc[NSB];
#pragma unroll
      for (int sb = 0; sb < NSB; ++sb) racc[sb] = f32x4_t{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int nj = 0; nj < DN; ++nj) {
#pragma unroll
        for (int kk = 0; kk < BN / 32; ++kk) {
          bf16x8_t vb =
              load_frag_b_trT_swz(vt_lds[cur], VP, nj * 16, kk * 32, lane);
#pragma unroll
          for (int sb = 0; sb < NSB; ++sb) {
            bf16x8_t pa = load_frag_a(pw + sb * 16 * VP, VP, kk * 32, lane);
            oacc[sb][nj] = MFMA_16x16x32(pa, vb, oacc[sb][nj]);
            if (nj == 0) racc[sb] = MFMA_16x16x32(pa, ones_frag, racc[sb]);
          }
        }
      }
#pragma unroll
      for (int sb = 0; sb < NSB; ++sb)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          l_i[sb][r] = l_i[sb][r] * alpha_s[sb][r] + racc[sb][r];
    }
    // stage tile jb+1 into the OTHER buffer (its previous contents —
    // tile jb-1 — were last read before the barrier that ended jb-1, so
    // no wave can still be reading it); T14: the global loads issued at
    // the top of this iteration have been landing under the MFMAs
    if (jb + 1 < nkb) write_lds(cur ^ 1);
    __syncthreads();  // tile jb reads done AND tile jb+1 writes visible
  }

  // ---- epilogue: O /= l (strided [s,b,h,d] store) + LSE ----
  bf16* Op = O + ((long)b * HQ + hq) * D;  // O contiguous [s, b, hq, d]
  const long sOs = (long)Bb * HQ * D;
  float* Lp = LSE + ((long)b * HQ + hq) * SQ;
#pragma unroll
  for (int sb = 0; sb < NSB; ++sb) {
    float inv_l[4];
#pragma unroll
    for (int r = 0; r < 4; ++r)
      inv_l[r] = (l_i[sb][r] > 0.f) ? 1.f / l_i[sb][r] : 0.f;
#pragma unroll
    for (int nj = 0; nj < DN; ++nj) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = qrow_w + sb * 16 + (lane >> 4) * 4 + r;
        if (qrow < SQ)
          Op[(long)qrow * sOs + nj * 16 + (lane & 15)] =
              f2bf(oacc[sb][nj][r] * inv_l[r]);
      }
    }
    if ((lane & 15) == 0) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = qrow_w + sb * 16 + (lane >> 4) * 4 + r;
        if (qrow < SQ)
          Lp[qrow] = m_i[sb][r] + __logf(fmaxf(l_i[sb][r], 1e-30f));
      }
    }
  }
}

extern "C" {
void launch_flash_fwd_dbuf(const void* q, const void* k, const void* v, void* o,
                      void* lse, int B, int HQ, int HKV, int SQ, int SKV,
                      int D, bool causal, float scale, int window,
                      const long* qstr, const long* kstr, const long* vstr,
                      hipStream_t stream) {
  // BM=128 when the BM=256 grid would leave CUs idle (TP=8: HQ_local=4)
  const bool small = ((long)((SQ + 255) / 256) * B * HQ) < 512;
  const int bm = small ? 128 : 256;
  dim3 grid((SQ + bm - 1) / bm, B * HQ);
  dim3 blk(512);
#define CASE(DD, CC)                                                          \
  do {                                                                        \
    if (small)                                                                \
      flash_fwd_dbuf_kernel<DD, CC, 1><<<grid, blk, 0, stream>>>(                  \
          (const bf16*)q, (const bf16*)k, (const bf16*)v, (bf16*)o,           \
          (float*)lse, SQ, SKV, B, HQ, HKV, scale, window, qstr[0], qstr[1],  \
          qstr[2], kstr[0], kstr[1], kstr[2], vstr[0], vstr[1], vstr[2]);     \
    else                                                                      \
      flash_fwd_dbuf_kernel<DD, CC, 2><<<grid, blk, 0, stream>>>(                  \
          (const bf16*)q, (const bf16*)k, (const bf16*)v, (bf16*)o,           \
          (float*)lse, SQ, SKV, B, HQ, HKV, scale, window, qstr[0], qstr[1],  \
          qstr[2], kstr[0], kstr[1], kstr[2], vstr[0], vstr[1], vstr[2]);     \
  } while (0)
  if (D == 128) {
    if (causal) CASE(128, true); else CASE(128, false);
  } else if (D == 64) {
    if (causal) CASE(64, true); else CASE(64, false);
  }
#undef CASE
}
}
