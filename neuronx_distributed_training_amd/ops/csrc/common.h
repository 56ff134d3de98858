// Common helpers for MI355X (gfx950, CDNA4) kernels.
// Wavefront = 64 lanes; block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DEVINL __device__ __forceinline__

using bf16 = __hip_bfloat16;
using bf16x2 = __hip_bfloat162;

// 8 bf16 = 16 B: one dwordx4 load per lane.
struct bf16x8 {
  bf16x2 a, b, c, d;
};
union Pack16B {
  int4 i4;
  bf16x8 h8;
  float4 f4;
};

DEVINL float bf2f(bf16 x) { return __bfloat162float(x); }
DEVINL bf16 f2bf(float x) { return __float2bfloat16(x); }

// wave-64 reduction (sum)
DEVINL float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}
DEVINL float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}

// block reduction over NT threads (NT multiple of 64, <= 1024)
template <int NT>
DEVINL float block_reduce_sum(float v, float* lds /* >= NT/64 floats */) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  constexpr int NW = NT / 64;
  if (wid == 0) {
    v = (lane < NW) ? lds[lane] : 0.f;
    v = wave_reduce_sum(v);
    if (lane == 0) lds[0] = v;
  }
  __syncthreads();
  float r = lds[0];
  __syncthreads();
  return r;
}

#define HIP_CHECK_LAST()                                            \
  do {                                                              \
    hipError_t e = hipGetLastError();                               \
    if (e != hipSuccess) {                                          \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e),       \
             __FILE__, __LINE__);                                   \
    }                                                               \
  } while (0)
