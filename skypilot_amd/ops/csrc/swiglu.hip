// Fused SwiGLU forward/backward for CDNA4 (gfx950).
//
// Input is the fused gate|up GEMM output gu [rows, 2M] (gate = first M
// columns): y = silu(g) * u.  Backward writes dgu [rows, 2M] in one
// pass, eliminating the eager silu/mul kernels and the backward
// torch.cat observed in the r01 profile.  Memory-bound; s16x8 loads.
#include "common.h"

__device__ __forceinline__ float sigmoidf_(float x) {
  return 1.f / (1.f + __expf(-x));
}

extern "C" __global__ void swiglu_fwd_kernel(
    const unsigned short* __restrict__ gu, unsigned short* __restrict__ y,
    long long rows, int M) {
  const int nvec = M / 8;
  const long long total = rows * nvec;
  const long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < total; i += stride) {
    long long row = i / nvec;
    int v = (int)(i % nvec);
    const s16x8 gv = *(const s16x8*)(gu + row * 2 * M + v * 8);
    const s16x8 uv = *(const s16x8*)(gu + row * 2 * M + M + v * 8);
    float out[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf2f((unsigned short)gv[j]);
      float u = bf2f((unsigned short)uv[j]);
      out[j] = g * sigmoidf_(g) * u;
    }
    *(s16x8*)(y + row * M + v * 8) = f32_to_bf8(out);
  }
}

extern "C" __global__ void swiglu_bwd_kernel(
    const unsigned short* __restrict__ gu,
    const unsigned short* __restrict__ dy,
    unsigned short* __restrict__ dgu, long long rows, int M) {
  const int nvec = M / 8;
  const long long total = rows * nvec;
  const long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < total; i += stride) {
    long long row = i / nvec;
    int v = (int)(i % nvec);
    const s16x8 gv = *(const s16x8*)(gu + row * 2 * M + v * 8);
    const s16x8 uv = *(const s16x8*)(gu + row * 2 * M + M + v * 8);
    const s16x8 dyv = *(const s16x8*)(dy + row * M + v * 8);
    float dg[8], du[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf2f((unsigned short)gv[j]);
      float u = bf2f((unsigned short)uv[j]);
      float d = bf2f((unsigned short)dyv[j]);
      float sg = sigmoidf_(g);
      float silu = g * sg;
      dg[j] = d * u * (sg + silu * (1.f - sg));
      du[j] = d * silu;
    }
    *(s16x8*)(dgu + row * 2 * M + v * 8) = f32_to_bf8(dg);
    *(s16x8*)(dgu + row * 2 * M + M + v * 8) = f32_to_bf8(du);
  }
}

extern "C" void swiglu_fwd_launch(const void* gu, void* y, long long rows,
                                  int M, hipStream_t stream) {
  int grid = membound_grid(rows * (M / 8), 256);
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const unsigned short*)gu, (unsigned short*)y, rows, M);
}

extern "C" void swiglu_bwd_launch(const void* gu, const void* dy, void* dgu,
                                  long long rows, int M,
                                  hipStream_t stream) {
  int grid = membound_grid(rows * (M / 8), 256);
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const unsigned short*)gu, (const unsigned short*)dy,
                     (unsigned short*)dgu, rows, M);
}
