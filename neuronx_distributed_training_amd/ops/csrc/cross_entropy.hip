// Fused vocab-parallel cross-entropy statistics for MI355X.
// Replaces the eager CE chain over [tokens, vocab/tp] logits (the
// reference's parallel_cross_entropy contract): one
// online-max+sum pass forward (no fp32 softmax materialization), one
// recompute pass backward. The TP all-reduces (max / sum-exp / target
// logit) stay in the Python wrapper (parallel/loss.py).
#include "common.h"

// logits [N, V] bf16 → per-row local max (f32), sumexp@max (f32),
// target logit (f32; 0 when this shard doesn't own the target).
template <int NT>
__global__ void ce_fwd_kernel(const bf16* __restrict__ logits,
                              const long* __restrict__ targets,
                              float* __restrict__ row_max,
                              float* __restrict__ row_sumexp,
                              float* __restrict__ tgt_logit, int V,
                              long vocab_start) {
  const int row = blockIdx.x;
  const bf16* lr = logits + (long)row * V;
  const int nvec = V >> 3;
  // online max+sum per thread
  float m = -1e30f, s = 0.f;
  for (int i = threadIdx.x; i < nvec; i += NT) {
    Pack16B p;
    p.i4 = ((const int4*)lr)[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      bf16x2 h = (&p.h8.a)[j];
      float a = bf2f(h.x), b = bf2f(h.y);
      float mn = fmaxf(m, fmaxf(a, b));
      s = s * __expf(m - mn) + __expf(a - mn) + __expf(b - mn);
      m = mn;
    }
  }
  // tail (V % 8)
  for (int i = (nvec << 3) + threadIdx.x; i < V; i += NT) {
    float a = bf2f(lr[i]);
    float mn = fmaxf(m, a);
    s = s * __expf(m - mn) + __expf(a - mn);
    m = mn;
  }
  // block-combine (m, s) pairs
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float mo = __shfl_down(m, off, 64);
    float so = __shfl_down(s, off, 64);
    float mn = fmaxf(m, mo);
    s = s * __expf(m - mn) + so * __expf(mo - mn);
    m = mn;
  }
  __shared__ float lm[NT / 64], ls[NT / 64];
  if (lane == 0) {
    lm[wid] = m;
    ls[wid] = s;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = lm[0], S = ls[0];
    for (int w = 1; w < NT / 64; ++w) {
      float mn = fmaxf(M, lm[w]);
      S = S * __expf(M - mn) + ls[w] * __expf(lm[w] - mn);
      M = mn;
    }
    row_max[row] = M;
    row_sumexp[row] = S;
    const long t = targets[row] - vocab_start;
    tgt_logit[row] = (t >= 0 && t < V) ? bf2f(lr[t]) : 0.f;
  }
}

// dlogits = (exp(x − gmax)/gsum − onehot) · gout   (bf16 out)
template <int NT>
__global__ void ce_bwd_kernel(const bf16* __restrict__ logits,
                              const long* __restrict__ targets,
                              const float* __restrict__ gmax,
                              const float* __restrict__ gsum,
                              const float* __restrict__ gout,
                              bf16* __restrict__ dlogits, int V,
                              long vocab_start) {
  const int row = blockIdx.x;
  const bf16* lr = logits + (long)row * V;
  bf16* dr = dlogits + (long)row * V;
  const float M = gmax[row];
  const float inv_s = 1.f / gsum[row];
  const float go = gout[row];
  const long t = targets[row] - vocab_start;
  const int nvec = V >> 3;
  for (int i = threadIdx.x; i < nvec; i += NT) {
    Pack16B p, o;
    p.i4 = ((const int4*)lr)[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      bf16x2 h = (&p.h8.a)[j];
      const long c0 = ((long)i << 3) + (j << 1);
      float g0 = __expf(bf2f(h.x) - M) * inv_s - (c0 == t ? 1.f : 0.f);
      float g1 = __expf(bf2f(h.y) - M) * inv_s - (c0 + 1 == t ? 1.f : 0.f);
      bf16x2 res;
      res.x = f2bf(g0 * go);
      res.y = f2bf(g1 * go);
      (&o.h8.a)[j] = res;
    }
    ((int4*)dr)[i] = o.i4;
  }
  for (int i = (nvec << 3) + threadIdx.x; i < V; i += NT) {
    float g = __expf(bf2f(lr[i]) - M) * inv_s - ((long)i == t ? 1.f : 0.f);
    dr[i] = f2bf(g * go);
  }
}

extern "C" {
void launch_ce_fwd(const void* logits, const void* targets, void* row_max,
                   void* row_sumexp, void* tgt_logit, long N, int V,
                   long vocab_start, hipStream_t stream) {
  ce_fwd_kernel<256><<<dim3((unsigned)N), dim3(256), 0, stream>>>(
      (const bf16*)logits, (const long*)targets, (float*)row_max,
      (float*)row_sumexp, (float*)tgt_logit, V, vocab_start);
}
void launch_ce_bwd(const void* logits, const void* targets, const void* gmax,
                   const void* gsum, const void* gout, void* dlogits, long N,
                   int V, long vocab_start, hipStream_t stream) {
  ce_bwd_kernel<256><<<dim3((unsigned)N), dim3(256), 0, stream>>>(
      (const bf16*)logits, (const long*)targets, (const float*)gmax,
      (const float*)gsum, (const float*)gout, (bf16*)dlogits, V, vocab_start);
}
}
