// Rotary position embedding (rotate-half), strided input → contiguous
// [s, b, h, d] output. Input is typically a non-contiguous [b, h, s, d]
// view of the fused QKV GEMM output — the kernel reads it in place, so the
// attention path does zero permute/contiguous copies.
// cos/sin tables [Smax, D] fp32; backward = forward with -sin (orthogonal).
#include "common.h"

__global__ void rope_fwd_kernel(const bf16* __restrict__ x,
                                const float* __restrict__ cost,
                                const float* __restrict__ sint,
                                bf16* __restrict__ y, int S, int B, int H,
                                int D, long sxs, long sxb, long sxh,
                                int pos_offset, int pos_offset2,
                                float sin_sign) {
  // pos_offset2 >= 0: zigzag context-parallel layout — rows [S/2, S) are
  // the rank's SECOND global chunk and take positions pos_offset2 + i
  const int half = D >> 1;
  const int shalf = S >> 1;
  const long total = (long)S * B * H * half;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int d = (int)(idx % half);
    const long row = idx / half;  // output row in [s][b][h] order
    const int h = (int)(row % H);
    const int b = (int)((row / H) % B);
    const int s = (int)(row / ((long)H * B));
    const bf16* xr = x + s * sxs + b * sxb + h * sxh;
    bf16* yr = y + row * D;
    const int p = (pos_offset2 >= 0 && s >= shalf)
                      ? (s - shalf) + pos_offset2
                      : s + pos_offset;
    const float c0 = cost[(long)p * D + d];
    const float s0 = sint[(long)p * D + d] * sin_sign;
    const float c1 = cost[(long)p * D + d + half];
    const float s1 = sint[(long)p * D + d + half] * sin_sign;
    const float x0 = bf2f(xr[d]);
    const float x1 = bf2f(xr[d + half]);
    yr[d] = f2bf(x0 * c0 - x1 * s0);
    yr[d + half] = f2bf(x1 * c1 + x0 * s1);
  }
}

extern "C" {
void launch_rope_fwd(const void* x, const void* cost, const void* sint,
                     void* y, int S, int B, int H, int D, const long* xstr,
                     int pos_offset, int pos_offset2, float sin_sign,
                     hipStream_t stream) {
  long total = (long)S * B * H * (D >> 1);
  int blocks = (int)min((total + 255) / 256, (long)16384);
  rope_fwd_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(
      (const bf16*)x, (const float*)cost, (const float*)sint, (bf16*)y, S, B,
      H, D, xstr[0], xstr[1], xstr[2], pos_offset, pos_offset2, sin_sign);
}
}
