// RMSNorm forward/backward for MI355X — one HBM pass each way.
// Replaces the reference's apex/NxD RMSNorm.
// bf16 IO, fp32 accumulate; vectorized 8×bf16 (16 B) loads per lane
// (guide: scalar bf16 loads are a 2× loss on this op).
#include "common.h"

// ---------------- forward ----------------
// x: [N, H] bf16, w: [H] bf16 -> y: [N, H] bf16, invrms: [N] f32.
// One block of 256 threads per row.
template <int NT>
__global__ void rmsnorm_fwd_kernel(const bf16* __restrict__ x,
                                   const bf16* __restrict__ w,
                                   bf16* __restrict__ y,
                                   float* __restrict__ invrms,
                                   int H, float eps) {
  __shared__ float lds[NT / 64];
  const int row = blockIdx.x;
  const bf16* xr = x + (long)row * H;
  bf16* yr = y + (long)row * H;
  float ss = 0.f;
  const int nvec = H >> 3;  // H % 8 == 0
  for (int i = threadIdx.x; i < nvec; i += NT) {
    Pack16B p;
    p.i4 = ((const int4*)xr)[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      bf16x2 h = (&p.h8.a)[j];
      float lo = bf2f(h.x), hi = bf2f(h.y);
      ss += lo * lo + hi * hi;
    }
  }
  ss = block_reduce_sum<NT>(ss, lds);
  float r = rsqrtf(ss / (float)H + eps);
  if (threadIdx.x == 0) invrms[row] = r;
  for (int i = threadIdx.x; i < nvec; i += NT) {
    Pack16B p, pw, o;
    p.i4 = ((const int4*)xr)[i];
    pw.i4 = ((const int4*)w)[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      bf16x2 h = (&p.h8.a)[j];
      bf16x2 hw = (&pw.h8.a)[j];
      bf16x2 res;
      res.x = f2bf(bf2f(h.x) * r * bf2f(hw.x));
      res.y = f2bf(bf2f(h.y) * r * bf2f(hw.y));
      (&o.h8.a)[j] = res;
    }
    ((int4*)yr)[i] = o.i4;
  }
}

// ---------------- backward ----------------
// dx = r*(w*dy - xhat * mean(w*dy*xhat));  dw = sum_rows(dy * xhat)
// Grid-stride over rows; per-block dw accumulated in LDS fp32 then one
// atomicAdd pass (contention = gridDim, not N).
template <int NT>
__global__ void rmsnorm_bwd_kernel(const bf16* __restrict__ dy,
                                   const bf16* __restrict__ x,
                                   const bf16* __restrict__ w,
                                   const float* __restrict__ invrms,
                                   bf16* __restrict__ dx,
                                   float* __restrict__ dw /* [H] f32, zeroed */,
                                   int N, int H) {
  extern __shared__ float smem[];          // [H] dw accum + [NT/64] reduce
  float* dw_lds = smem;
  float* red = smem + H;
  for (int i = threadIdx.x; i < H; i += NT) dw_lds[i] = 0.f;
  __syncthreads();

  const int nvec = H >> 3;
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const bf16* dyr = dy + (long)row * H;
    const bf16* xr = x + (long)row * H;
    bf16* dxr = dx + (long)row * H;
    const float r = invrms[row];
    // pass 1: c = mean(w*dy*xhat)
    float c = 0.f;
    for (int i = threadIdx.x; i < nvec; i += NT) {
      Pack16B pd, px, pw;
      pd.i4 = ((const int4*)dyr)[i];
      px.i4 = ((const int4*)xr)[i];
      pw.i4 = ((const int4*)w)[i];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        bf16x2 hd = (&pd.h8.a)[j], hx = (&px.h8.a)[j], hw = (&pw.h8.a)[j];
        c += bf2f(hw.x) * bf2f(hd.x) * bf2f(hx.x) * r;
        c += bf2f(hw.y) * bf2f(hd.y) * bf2f(hx.y) * r;
      }
    }
    c = block_reduce_sum<NT>(c, red) / (float)H;
    // pass 2: dx store + dw partial accumulation (LDS).
    // Each thread owns fixed columns (i strided by NT) so the dw_lds adds
    // are conflict-free plain adds, no atomics needed within a block.
    for (int i = threadIdx.x; i < nvec; i += NT) {
      Pack16B pd, px, pw, o;
      pd.i4 = ((const int4*)dyr)[i];
      px.i4 = ((const int4*)xr)[i];
      pw.i4 = ((const int4*)w)[i];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        bf16x2 hd = (&pd.h8.a)[j], hx = (&px.h8.a)[j], hw = (&pw.h8.a)[j];
        float xh0 = bf2f(hx.x) * r, xh1 = bf2f(hx.y) * r;
        float d0 = bf2f(hd.x), d1 = bf2f(hd.y);
        bf16x2 res;
        res.x = f2bf((bf2f(hw.x) * d0 - xh0 * c) * r);
        res.y = f2bf((bf2f(hw.y) * d1 - xh1 * c) * r);
        (&o.h8.a)[j] = res;
        const int col = (i << 3) + (j << 1);
        dw_lds[col] += d0 * xh0;
        dw_lds[col + 1] += d1 * xh1;
      }
      ((int4*)dxr)[i] = o.i4;
    }
  }
  __syncthreads();
  // flush dw partials
  for (int i = threadIdx.x; i < H; i += NT) atomicAdd(&dw[i], dw_lds[i]);
}

extern "C" {
void launch_rmsnorm_fwd(const void* x, const void* w, void* y, void* invrms,
                        long N, int H, float eps, hipStream_t stream) {
  constexpr int NT = 256;
  rmsnorm_fwd_kernel<NT><<<dim3((unsigned)N), dim3(NT), 0, stream>>>(
      (const bf16*)x, (const bf16*)w, (bf16*)y, (float*)invrms, H, eps);
}

void launch_rmsnorm_bwd(const void* dy, const void* x, const void* w,
                        const void* invrms, void* dx, void* dw, long N, int H,
                        hipStream_t stream) {
  constexpr int NT = 256;
  int blocks = (int)min((long)1024, N);
  size_t shmem = (H + NT / 64) * sizeof(float);
  rmsnorm_bwd_kernel<NT><<<dim3(blocks), dim3(NT), shmem, stream>>>(
      (const bf16*)dy, (const bf16*)x, (const bf16*)w, (const float*)invrms,
      (bf16*)dx, (float*)dw, (int)N, H);
}
}
