// MFMA 16x16x32 bf16 layout probe — single wave, verifies the fragment
// layouts documented in common.h against a CPU A@B reference (asymmetric
// inputs; see guide §3 "Always A=I-check with ASYMMETRIC B").
#include "common.h"

extern "C" __global__ void mfma_probe_kernel(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int lrow = lane & 15;
  const int lgrp = lane >> 4;
  s16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (short)A[lrow * 32 + lgrp * 8 + j];      // A[l&15][(l>>4)*8+j]
    b[j] = (short)B[(lgrp * 8 + j) * 16 + lrow];    // B[(l>>4)*8+j][l&15]
  }
  f32x4 c = f32x4{0.f, 0.f, 0.f, 0.f};
  c = MFMA_BF16(as_bf16x8(a), as_bf16x8(b), c);
#pragma unroll
  for (int r = 0; r < 4; ++r) C[(lgrp * 4 + r) * 16 + lrow] = c[r];
}

extern "C" void mfma_probe_launch(const void* A, const void* B, float* C,
                                  hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const unsigned short*)A, (const unsigned short*)B, C);
}
