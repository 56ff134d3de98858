#include "hip/hip_runtime.h"
// Fused SwiGLU fwd/bwd: y = silu(gate) * up with gate_up packed
// [N, 2I] = [gate | up] (fused ColumnParallel output). One HBM pass.
#include "common.h"

__global__ void swiglu_fwd_kernel(const bf16* __restrict__ gu,
                                  bf16* __restrict__ y, long N, int I) {
  const int nvec = I >> 3;
  const long total = N * (long)nvec;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / nvec;
    const int i = (int)(idx % nvec);
    const bf16* g = gu + row * 2 * (long)I;
    const bf16* u = g + I;
    Pack16B pg, pu, o;
    pg.i4 = ((const int4*)g)[i];
    pu.i4 = ((const int4*)u)[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      bf16x2 hg = (&pg.h8.a)[j], hu = (&pu.h8.a)[j];
      float g0 = bf2f(hg.x), g1 = bf2f(hg.y);
      float s0 = g0 / (1.f + __expf(-g0));
      float s1 = g1 / (1.f + __expf(-g1));
      bf16x2 res;
      res.x = f2bf(s0 * bf2f(hu.x));
      res.y = f2bf(s1 * bf2f(hu.y));
      (&o.h8.a)[j] = res;
    }
    ((int4*)(y + row * (long)I))[i] = o.i4;
  }
}

__global__ void swiglu_bwd_kernel(const bf16* __restrict__ dy,
                                  const bf16* __restrict__ gu,
                                  bf16* __restrict__ dgu, long N, int I) {
  const int nvec = I >> 3;
  const long total = N * (long)nvec;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / nvec;
    const int i = (int)(idx % nvec);
    const bf16* g = gu + row * 2 * (long)I;
    const bf16* u = g + I;
    bf16* dg = dgu + row * 2 * (long)I;
    bf16* du = dg + I;
    Pack16B pg, pu, pd, og, ou;
    pg.i4 = ((const int4*)g)[i];
    pu.i4 = ((const int4*)u)[i];
    pd.i4 = ((const int4*)(dy + row * (long)I))[i];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      bf16x2 hg = (&pg.h8.a)[j], hu = (&pu.h8.a)[j], hd = (&pd.h8.a)[j];
      float g0 = bf2f(hg.x), g1 = bf2f(hg.y);
      float u0 = bf2f(hu.x), u1 = bf2f(hu.y);
      float d0 = bf2f(hd.x), d1 = bf2f(hd.y);
      float sig0 = 1.f / (1.f + __expf(-g0));
      float sig1 = 1.f / (1.f + __expf(-g1));
      float silu0 = g0 * sig0, silu1 = g1 * sig1;
      float ds0 = sig0 * (1.f + g0 * (1.f - sig0));
      float ds1 = sig1 * (1.f + g1 * (1.f - sig1));
      bf16x2 rg, ru;
      rg.x = f2bf(d0 * u0 * ds0);
      rg.y = f2bf(d1 * u1 * ds1);
      ru.x = f2bf(d0 * silu0);
      ru.y = f2bf(d1 * silu1);
      (&og.h8.a)[j] = rg;
      (&ou.h8.a)[j] = ru;
    }
    ((int4*)dg)[i] = og.i4;
    ((int4*)du)[i] = ou.i4;
  }
}

extern "C" {
void launch_swiglu_fwd(const void* gu, void* y, long N, int I,
                       hipStream_t stream) {
  long total = N * (long)(I >> 3);
  int blocks = (int)min((total + 255) / 256, (long)8192);
 hipLaunchKernelGGL(( swiglu_fwd_kernel), dim3(dim3(blocks)), dim3(dim3(256)), 0, stream, (const bf16*)gu,
                                                            (bf16*)y, N, I);
}
void launch_swiglu_bwd(const void* dy, const void* gu, void* dgu, long N,
                       int I, hipStream_t stream) {
  long total = N * (long)(I >> 3);
  int blocks = (int)min((total + 255) / 256, (long)8192);
 hipLaunchKernelGGL(( swiglu_bwd_kernel), dim3(dim3(blocks)), dim3(dim3(256)), 0, stream, 
      (const bf16*)dy, (const bf16*)gu, (bf16*)dgu, N, I);
}
}
