// Fused AdamW on the ZeRO-1 fp32 shard (master weights + m/v state),
// with a per-element decoupled-weight-decay mask. One fused HBM pass over
// 4 fp32 streams replaces ~8 separate eager kernels.
//
// Round-2 extensions (kill the optimizer data-motion hotspot — the
// 85 ms elementwise passes around the flat buffer):
//   * grad_scale: device-scalar pointer multiplied into the grad on load —
//     folds the grad-clip (and any DP divisor) into this pass, removing a
//     separate 64 GB read-modify-write over the fp32 grad buffer.
//   * bf16_out: optional bf16 destination (the model-dtype param_flat
//     shard) written alongside the fp32 master — removes the separate
//     master->bf16 cast pass AND the bf16 param copy pass.
#include "common.h"

__global__ void adamw_kernel(float* __restrict__ p, const float* __restrict__ g,
                             float* __restrict__ m, float* __restrict__ v,
                             const bool* __restrict__ wd_mask,
                             const float* __restrict__ grad_scale,
                             bf16* __restrict__ p_bf16, long n, float lr,
                             float beta1, float beta2, float eps, float wd,
                             float bc1, float bc2) {
  const float step_size = lr / bc1;
  const float gscale = grad_scale ? *grad_scale : 1.f;
  for (long i = (blockIdx.x * (long)blockDim.x + threadIdx.x) * 4; i < n;
       i += (long)gridDim.x * blockDim.x * 4) {
    // 16-B vector path when 4 elements remain
    if (i + 4 <= n) {
      float4 pv = *(const float4*)(p + i);
      float4 gv = *(const float4*)(g + i);
      float4 mv = *(const float4*)(m + i);
      float4 vv = *(const float4*)(v + i);
      float* pp = &pv.x;
      const float* gg = &gv.x;
      float* mm = &mv.x;
      float* vs = &vv.x;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float grad = gg[j] * gscale;
        mm[j] = beta1 * mm[j] + (1.f - beta1) * grad;
        vs[j] = beta2 * vs[j] + (1.f - beta2) * grad * grad;
        float denom = sqrtf(vs[j] / bc2) + eps;
        float decay = wd_mask[i + j] ? (1.f - lr * wd) : 1.f;
        pp[j] = pp[j] * decay - step_size * mm[j] / denom;
      }
      *(float4*)(p + i) = pv;
      *(float4*)(m + i) = mv;
      *(float4*)(v + i) = vv;
      if (p_bf16) {
        bf16x2 h0{f2bf(pv.x), f2bf(pv.y)};
        bf16x2 h1{f2bf(pv.z), f2bf(pv.w)};
        // 8-B store of 4 bf16
        union { bf16x2 h[2]; int2 i2; } u{{h0, h1}};
        *(int2*)(p_bf16 + i) = u.i2;
      }
    } else {
      for (long j = i; j < n; ++j) {
        float grad = g[j] * gscale;
        m[j] = beta1 * m[j] + (1.f - beta1) * grad;
        v[j] = beta2 * v[j] + (1.f - beta2) * grad * grad;
        float denom = sqrtf(v[j] / bc2) + eps;
        float decay = wd_mask[j] ? (1.f - lr * wd) : 1.f;
        p[j] = p[j] * decay - step_size * m[j] / denom;
        if (p_bf16) p_bf16[j] = f2bf(p[j]);
      }
    }
  }
}

extern "C" {
void launch_adamw(void* p, const void* g, void* m, void* v,
                  const void* wd_mask, const void* grad_scale, void* p_bf16,
                  long n, float lr, float beta1, float beta2, float eps,
                  float wd, int step, hipStream_t stream) {
  float bc1 = 1.f - powf(beta1, (float)step);
  float bc2 = 1.f - powf(beta2, (float)step);
  long quads = (n + 3) / 4;
  int blocks = (int)min((quads + 255) / 256, (long)8192);
  adamw_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(
      (float*)p, (const float*)g, (float*)m, (float*)v, (const bool*)wd_mask,
      (const float*)grad_scale, (bf16*)p_bf16, n, lr, beta1, beta2, eps, wd,
      bc1, bc2);
}
}
