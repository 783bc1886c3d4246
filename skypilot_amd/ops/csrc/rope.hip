// Rotary position embedding (Llama half-rotation style) for CDNA4.
//
// Trig tables are precomputed on the host side (guide App. B: on-device
// sinf/cosf turns a memory-bound op VALU-bound) and passed as fp32
// cos/sin [S, D/2] device buffers.
//
// Layout: x is [T, Hq, D] contiguous (T = B*S tokens), positions[T] gives
// each token's table row. One wave per (token, head): lane i handles the
// rotation pair (i, i + D/2), D <= 128.  backward == forward with -sin.
#include "common.h"

extern "C" __global__ void rope_kernel(
    const unsigned short* __restrict__ x, unsigned short* __restrict__ y,
    const float* __restrict__ cos_tab, const float* __restrict__ sin_tab,
    const int* __restrict__ positions, long long n_tokens, int n_heads, int D,
    float sin_sign) {
  const int half = D / 2;
  const long long rows = n_tokens * n_heads;  // one row = one head vector
  const int lane = threadIdx.x & (WAVE - 1);
  const long long wave_id =
      (long long)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  const long long stride = (long long)gridDim.x * (blockDim.x / WAVE);

  for (long long row = wave_id; row < rows; row += stride) {
    const long long tok = row / n_heads;
    const int pos = positions[tok];
    const unsigned short* xr = x + row * D;
    unsigned short* yr = y + row * D;
    for (int i = lane; i < half; i += WAVE) {
      float c = cos_tab[(long long)pos * half + i];
      float s = sin_tab[(long long)pos * half + i] * sin_sign;
      float x1 = bf2f(xr[i]);
      float x2 = bf2f(xr[i + half]);
      yr[i] = f2bf(x1 * c - x2 * s);
      yr[i + half] = f2bf(x2 * c + x1 * s);
    }
  }
}

extern "C" void rope_launch(const void* x, void* y, const float* cos_tab,
                            const float* sin_tab, const int* positions,
                            long long n_tokens, int n_heads, int D,
                            bool backward, hipStream_t stream) {
  long long waves = n_tokens * n_heads;
  int waves_per_block = 4;  // 256 threads
  int grid = membound_grid(waves, waves_per_block);
  hipLaunchKernelGGL(rope_kernel, dim3(grid), dim3(waves_per_block * WAVE), 0,
                     stream, (const unsigned short*)x, (unsigned short*)y,
                     cos_tab, sin_tab, positions, n_tokens, n_heads, D,
                     backward ? -1.f : 1.f);
}
