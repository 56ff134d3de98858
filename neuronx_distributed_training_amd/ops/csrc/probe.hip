// MFMA layout probe: one wave computes C[16,16] = A[16,32] @ B[32,16]
// using the fragment maps in attn_common.h. The GPU test compares against
// torch.matmul — a failure means the lane→element maps are wrong and every
// attention kernel is too (asymmetric B catches transposes).
#include "attn_common.h"

__global__ void mfma_probe_kernel(const bf16* __restrict__ A,
                                  const bf16* __restrict__ B,
                                  float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  bf16x8_t af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = (__bf16)A[(lane & 15) * 32 + (lane >> 4) * 8 + j];
    bf[j] = (__bf16)B[((lane >> 4) * 8 + j) * 16 + (lane & 15)];
  }
  f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
  acc = MFMA_16x16x32(af, bf, acc);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    C[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

extern "C" void launch_mfma_probe(const void* a, const void* b, void* c,
                                  hipStream_t stream) {
  mfma_probe_kernel<<<1, 64, 0, stream>>>((const bf16*)a, (const bf16*)b,
                                          (float*)c);
}

// 32×32×16 probe for the v3 (T12 swapped-operand) attention design:
// assumed maps (to be confirmed on HW before v3 lands — ROADMAP.md §1):
//   A (32×16): row = lane&31, k = (lane>>5)*8 + j     (8 elems, 4 VGPRs)
//   B (16×32): col = lane&31, k = (lane>>5)*8 + j
//   C (32×32): col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
typedef float f32x16_t __attribute__((ext_vector_type(16)));

__global__ void mfma_probe32_kernel(const bf16* __restrict__ A,
                                    const bf16* __restrict__ B,
                                    float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  bf16x8_t af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = (__bf16)A[(lane & 31) * 16 + (lane >> 5) * 8 + j];
    bf[j] = (__bf16)B[((lane >> 5) * 8 + j) * 32 + (lane & 31)];
  }
  f32x16_t acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    C[row * 32 + (lane & 31)] = acc[r];
  }
}

extern "C" void launch_mfma_probe32(const void* a, const void* b, void* c,
                                    hipStream_t stream) {
  mfma_probe32_kernel<<<1, 64, 0, stream>>>((const bf16*)a, (const bf16*)b,
                                            (float*)c);
}
