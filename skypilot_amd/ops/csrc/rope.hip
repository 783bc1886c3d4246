// Rotary position embedding (Llama half-rotation style) for CDNA4.
//
// Trig tables are precomputed on the host side (guide App. B: on-device
// sinf/cosf turns a memory-bound op VALU-bound) and passed as fp32
// cos/sin [S, D/2] device buffers.
//
// Layout: x is [T, H, D] contiguous (T = B*S tokens), positions[T] gives
// each token's table row.  Vectorized path (D % 32 == 0): 16 lanes per
// (token, head) row, each lane rotates 4 pairs with 8B bf16 loads and
// 16B table loads (guide §6 G13: scalar bf16 costs ~2-2.5x).
// backward == forward with -sin.
#include "common.h"

extern "C" __global__ void rope_kernel_vec(
    const unsigned short* __restrict__ x, unsigned short* __restrict__ y,
    const float* __restrict__ cos_tab, const float* __restrict__ sin_tab,
    const int* __restrict__ positions, long long n_tokens, int n_heads,
    int D, float sin_sign) {
  const int half = D / 2;
  const int lanes_per_row = 16;
  const int pairs_per_lane = half / lanes_per_row;  // 4 for D=128
  const long long rows = n_tokens * n_heads;
  const int sub = threadIdx.x & 15;          // lane-within-row
  const long long row0 =
      ((long long)blockIdx.x * blockDim.x + threadIdx.x) >> 4;
  const long long stride = ((long long)gridDim.x * blockDim.x) >> 4;

  for (long long row = row0; row < rows; row += stride) {
    const long long tok = row / n_heads;
    const int pos = positions[tok];
    const unsigned short* xr = x + row * D;
    unsigned short* yr = y + row * D;
    const float* ct = cos_tab + (long long)pos * half + sub * pairs_per_lane;
    const float* st = sin_tab + (long long)pos * half + sub * pairs_per_lane;
#pragma unroll 2
    for (int p = 0; p < pairs_per_lane; p += 4) {
      const int i = sub * pairs_per_lane + p;
      s16x4 x1 = *(const s16x4*)(xr + i);
      s16x4 x2 = *(const s16x4*)(xr + i + half);
      f32x4 c = *(const f32x4*)(ct + p);
      f32x4 sn = *(const f32x4*)(st + p);
      s16x4 o1, o2;
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        float a = bf2f((unsigned short)x1[k]);
        float b2 = bf2f((unsigned short)x2[k]);
        float sk = sn[k] * sin_sign;
        o1[k] = (short)f2bf(a * c[k] - b2 * sk);
        o2[k] = (short)f2bf(b2 * c[k] + a * sk);
      }
      *(s16x4*)(yr + i) = o1;
      *(s16x4*)(yr + i + half) = o2;
    }
  }
}

// Scalar fallback for head dims not divisible by 32.
extern "C" __global__ void rope_kernel(
    const unsigned short* __restrict__ x, unsigned short* __restrict__ y,
    const float* __restrict__ cos_tab, const float* __restrict__ sin_tab,
    const int* __restrict__ positions, long long n_tokens, int n_heads, int D,
    float sin_sign) {
  const int half = D / 2;
  const long long rows = n_tokens * n_heads;
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
  const float ss = backward ? -1.f : 1.f;
  if (D % 32 == 0) {
    long long work = n_tokens * n_heads * 16;  // 16 lanes per row
    int grid = membound_grid(work, 256);
    hipLaunchKernelGGL(rope_kernel_vec, dim3(grid), dim3(256), 0, stream,
                       (const unsigned short*)x, (unsigned short*)y, cos_tab,
                       sin_tab, positions, n_tokens, n_heads, D, ss);
    return;
  }
  long long waves = n_tokens * n_heads;
  int grid = membound_grid(waves, 4);
  hipLaunchKernelGGL(rope_kernel, dim3(grid), dim3(4 * WAVE), 0, stream,
                     (const unsigned short*)x, (unsigned short*)y, cos_tab,
                     sin_tab, positions, n_tokens, n_heads, D, ss);
}
