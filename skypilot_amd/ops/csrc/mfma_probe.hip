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

// 32x32x16 bf16 probe — layouts (verified by tests/test_gpu_ops.py):
//   A[32x16]: lane l holds A[l&31][(l>>5)*8 + j]          j=0..7
//   B[16x32]: lane l holds B[(l>>5)*8 + j][l&31]
//   C/D[32x32]: lane l reg r holds C[(r&3)+8*(r>>2)+4*(l>>5)][l&31]
typedef __attribute__((ext_vector_type(16))) float f32x16;
#define MFMA32_BF16(a, b, c) \
  __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0)

extern "C" __global__ void mfma32_probe_kernel(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int col = lane & 31;
  const int half = lane >> 5;
  s16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (short)A[col * 16 + half * 8 + j];       // A[l&31][(l>>5)*8+j]
    b[j] = (short)B[(half * 8 + j) * 32 + col];     // B[(l>>5)*8+j][l&31]
  }
  f32x16 c;
#pragma unroll
  for (int r = 0; r < 16; ++r) c[r] = 0.f;
  c = MFMA32_BF16(as_bf16x8(a), as_bf16x8(b), c);
#pragma unroll
  for (int r = 0; r < 16; ++r)
    C[((r & 3) + 8 * (r >> 2) + 4 * half) * 32 + col] = c[r];
}

extern "C" void mfma32_probe_launch(const void* A, const void* B, float* C,
                                    hipStream_t stream) {
  hipLaunchKernelGGL(mfma32_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const unsigned short*)A, (const unsigned short*)B, C);
}

// ds_read_b64_tr_b16 semantics probe (round-2 groundwork, guide T10):
// LDS is filled with bf16 value == element index; each lane issues one
// tr-read at addr = base + lane*8 (contiguous per-lane, as a plain b64
// read would) and we record which 4 elements each lane received.
extern "C" __global__ void trb16_probe_kernel(float* __restrict__ out,
                                              int mode) {
  __shared__ unsigned short buf[1024];
  const int lane = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 1024; i += blockDim.x)
    buf[i] = f2bf((float)(i + 100));  // distinguishable from zero
  __syncthreads();
  unsigned int base = (unsigned int)(unsigned long long)&buf[0];
  unsigned int addr;
  if (mode == 0)
    addr = base + lane * 8;            // contiguous 8B per lane
  else if (mode == 1)
    addr = base;                       // uniform base
  else
    addr = base + ((lane & 15) + (lane >> 4) * 64) * 2;
  unsigned long long dst;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %1\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=v"(dst)
      : "v"(addr)
      : "memory");
  __builtin_amdgcn_sched_barrier(0);
  unsigned short e[4] = {(unsigned short)(dst & 0xffff),
                         (unsigned short)((dst >> 16) & 0xffff),
                         (unsigned short)((dst >> 32) & 0xffff),
                         (unsigned short)((dst >> 48) & 0xffff)};
#pragma unroll
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = bf2f(e[j]);
}

extern "C" void trb16_probe_launch(float* out, int mode,
                                   hipStream_t stream) {
  hipLaunchKernelGGL(trb16_probe_kernel, dim3(1), dim3(64), 0, stream, out,
                     mode);
}

// permlane32_swap semantics probe (round-2 swapped-QK^T groundwork).
extern "C" __global__ void permlane_probe_kernel(float* __restrict__ out) {
  const int lane = threadIdx.x & 63;
  unsigned int a = 1000 + lane;   // marker values
  unsigned int b = 2000 + lane;
  auto pair = __builtin_amdgcn_permlane32_swap(a, b, false, false);
  out[lane * 2 + 0] = (float)pair[0];
  out[lane * 2 + 1] = (float)pair[1];
}

extern "C" void permlane_probe_launch(float* out, hipStream_t stream) {
  hipLaunchKernelGGL(permlane_probe_kernel, dim3(1), dim3(64), 0, stream,
                     out);
}
