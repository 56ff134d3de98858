// Rotary position embedding (rotate-half), fused over [B, H, S, D].
// cos/sin tables [Smax, D] fp32; backward = forward with -sin (orthogonal).
#include "common.h"

// Each thread handles 2 paired elements (d, d + D/2) of one row; rows are
// (B*H*S); vectorized float2-on-bf16 via bf16x2 when D/2 % 2 == 0.
__global__ void rope_fwd_kernel(const bf16* __restrict__ x,
                                const float* __restrict__ cost,
                                const float* __restrict__ sint,
                                bf16* __restrict__ y, long BH, int S, int D,
                                int pos_offset, float sin_sign) {
  const int half = D >> 1;
  const long total = BH * (long)S * half;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const int d = (int)(idx % half);
    const long row = idx / half;          // (bh, s)
    const int s = (int)(row % S);
    const long base = row * D;
    const int p = s + pos_offset;
    float c0 = cost[(long)p * D + d];
    float s0 = sint[(long)p * D + d] * sin_sign;
    float c1 = cost[(long)p * D + d + half];
    float s1 = sint[(long)p * D + d + half] * sin_sign;
    float x0 = bf2f(x[base + d]);
    float x1 = bf2f(x[base + d + half]);
    // y0 = x0*c0 - x1*s0 ; y1 = x1*c1 + x0*s1  (rotate_half convention)
    y[base + d] = f2bf(x0 * c0 - x1 * s0);
    y[base + d + half] = f2bf(x1 * c1 + x0 * s1);
  }
}

extern "C" {
void launch_rope_fwd(const void* x, const void* cost, const void* sint,
                     void* y, long BH, int S, int D, int pos_offset,
                     float sin_sign, hipStream_t stream) {
  long total = BH * (long)S * (D >> 1);
  int blocks = (int)min((total + 255) / 256, (long)16384);
  rope_fwd_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(
      (const bf16*)x, (const float*)cost, (const float*)sint, (bf16*)y, BH, S,
      D, pos_offset, sin_sign);
}
}
