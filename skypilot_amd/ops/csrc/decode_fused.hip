// Decode-step fusions.  The graphed decode forward is a chain of
// ~4.5 us-floor kernels (rocprofv3: rope x2 + cache-scatter x2 +
// residual adds + norms = ~1.4 ms of a 3.7 ms 8B step); these two
// kernels collapse 16 launches/layer to 9:
//
//   rmsnorm_res : x2 = x + res;  h = rmsnorm(x2) * w   (one kernel for
//                 the residual add AND the next norm; emits both)
//   rope_kvwrite: consumes the PACKED qkv GEMV output [n,(Hq+2Hkv)*D],
//                 applies rope to q and k, scatters k/v rows straight
//                 into the KV cache, emits contiguous roped q —
//                 replacing rope(q), rope(k), index_put(k), index_put(v)
//                 and the qkv split copies.
//
// Decode/serving only (no autograd).  Conventions match rope.hip
// (half-rotation, fp32 cos/sin [S, D/2]) and kv_cache.py
// ([slots, S_max, Hkv, D] per layer).
#include "common.h"

extern "C" __global__ void rmsnorm_res_kernel(
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ res,
    const unsigned short* __restrict__ w,
    unsigned short* __restrict__ x_out,
    unsigned short* __restrict__ h_out, int H, float eps) {
  __shared__ float red[16];
  const long long row = blockIdx.x;
  const unsigned short* xr = x + row * H;
  const unsigned short* rr = res + row * H;
  float vals[32];  // H <= 8192, 256 threads
  float ss = 0.f;
  if (H % 2048 == 0) {
    // vector path: b128 loads/stores (scalar bf16 costs 2-2.5x, G13)
    const int nv = H / 2048;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      if (c >= nv) break;
      int i = c * 2048 + threadIdx.x * 8;
      s16x8 xv = *(const s16x8*)(xr + i);
      s16x8 rv = *(const s16x8*)(rr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf2f((unsigned short)xv[j]) +
                  bf2f((unsigned short)rv[j]);
        vals[c * 8 + j] = v;
        ss += v * v;
      }
    }
    const float inv = rsqrtf(block_reduce_sum(ss, red) / H + eps);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      if (c >= nv) break;
      int i = c * 2048 + threadIdx.x * 8;
      s16x8 wv = *(const s16x8*)(w + i);
      s16x8 xo, ho;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned short xb = f2bf(vals[c * 8 + j]);
        xo[j] = (short)xb;
        // match rmsnorm.hip: normalize the bf16-rounded sum
        ho[j] = (short)f2bf(bf2f(xb) * inv *
                            bf2f((unsigned short)wv[j]));
      }
      *(s16x8*)(x_out + row * H + i) = xo;
      *(s16x8*)(h_out + row * H + i) = ho;
    }
    return;
  }
  const int nchunk = H / 256;
  // constant trip bound so `vals` fully promotes to registers
#pragma unroll
  for (int c = 0; c < 32; ++c) {
    if (c >= nchunk) break;
    int i = c * 256 + threadIdx.x;
    float v = bf2f(xr[i]) + bf2f(rr[i]);
    vals[c] = v;
    ss += v * v;
  }
  const float inv = rsqrtf(block_reduce_sum(ss, red) / H + eps);
#pragma unroll
  for (int c = 0; c < 32; ++c) {
    if (c >= nchunk) break;
    int i = c * 256 + threadIdx.x;
    unsigned short xv = f2bf(vals[c]);
    x_out[row * H + i] = xv;
    h_out[row * H + i] = f2bf(bf2f(xv) * inv * bf2f(w[i]));
  }
}

extern "C" __global__ void rope_kvwrite_kernel(
    const unsigned short* __restrict__ qkv,  // [n, (Hq+2*Hkv)*D]
    unsigned short* __restrict__ q_out,      // [n, Hq*D]
    unsigned short* __restrict__ kc,         // [slots, S_max, Hkv, D]
    unsigned short* __restrict__ vc,
    const float* __restrict__ cos_tab, const float* __restrict__ sin_tab,
    const int* __restrict__ positions,       // [n] rope row == write row
    const int* __restrict__ slot_ids,        // [n]
    int n, int Hq, int Hkv, int D, int S_max) {
  const int half = D / 2;
  const int pairs = half / 16;               // lanes_per_row = 16
  const int heads = Hq + 2 * Hkv;
  const long long rows = (long long)n * heads;
  const int sub = threadIdx.x & 15;
  const long long row0 =
      ((long long)blockIdx.x * blockDim.x + threadIdx.x) >> 4;
  const long long stride = ((long long)gridDim.x * blockDim.x) >> 4;
  for (long long row = row0; row < rows; row += stride) {
    const int tok = (int)(row / heads);
    const int j = (int)(row % heads);
    const int p = positions[tok];
    const unsigned short* src = qkv + row * D;
    unsigned short* dst;
    bool rot;
    if (j < Hq) {                    // roped q, stays contiguous
      dst = q_out + ((long long)tok * Hq + j) * D;
      rot = true;
    } else if (j < Hq + Hkv) {       // roped k -> cache row
      dst = kc + (((long long)slot_ids[tok] * S_max + p) * Hkv +
                  (j - Hq)) * D;
      rot = true;
    } else {                         // v -> cache row, no rotation
      dst = vc + (((long long)slot_ids[tok] * S_max + p) * Hkv +
                  (j - Hq - Hkv)) * D;
      rot = false;
    }
    if (!rot) {
      const int i = sub * pairs;     // pairs elems from each half
      *(s16x4*)(dst + i) = *(const s16x4*)(src + i);
      *(s16x4*)(dst + i + half) = *(const s16x4*)(src + i + half);
      continue;
    }
    const float* ct = cos_tab + (long long)p * half + sub * pairs;
    const float* st = sin_tab + (long long)p * half + sub * pairs;
    const int i = sub * pairs;
    s16x4 x1 = *(const s16x4*)(src + i);
    s16x4 x2 = *(const s16x4*)(src + i + half);
    f32x4 c = *(const f32x4*)(ct);
    f32x4 sn = *(const f32x4*)(st);
    s16x4 o1, o2;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float a = bf2f((unsigned short)x1[k]);
      float b = bf2f((unsigned short)x2[k]);
      o1[k] = (short)f2bf(a * c[k] - b * sn[k]);
      o2[k] = (short)f2bf(b * c[k] + a * sn[k]);
    }
    *(s16x4*)(dst + i) = o1;
    *(s16x4*)(dst + i + half) = o2;
  }
}

extern "C" void rmsnorm_res_launch(const void* x, const void* res,
                                   const void* w, void* x_out, void* h_out,
                                   int rows, int H, float eps,
                                   hipStream_t stream) {
  hipLaunchKernelGGL(rmsnorm_res_kernel, dim3(rows), dim3(256), 0, stream,
                     (const unsigned short*)x, (const unsigned short*)res,
                     (const unsigned short*)w, (unsigned short*)x_out,
                     (unsigned short*)h_out, H, eps);
}

extern "C" void rope_kvwrite_launch(const void* qkv, void* q_out, void* kc,
                                    void* vc, const void* cos_tab,
                                    const void* sin_tab,
                                    const void* positions,
                                    const void* slot_ids, int n, int Hq,
                                    int Hkv, int D, int S_max,
                                    hipStream_t stream) {
  long long rows = (long long)n * (Hq + 2 * Hkv);
  int blocks = (int)((rows * 16 + 255) / 256);
  hipLaunchKernelGGL(rope_kvwrite_kernel, dim3(blocks), dim3(256), 0,
                     stream, (const unsigned short*)qkv,
                     (unsigned short*)q_out, (unsigned short*)kc,
                     (unsigned short*)vc, (const float*)cos_tab,
                     (const float*)sin_tab, (const int*)positions,
                     (const int*)slot_ids, n, Hq, Hkv, D, S_max);
}

// Fused greedy-decode advance: per-row argmax over the logits + the
// whole in-graph state bump (ring token store, next-token stage,
// positions/pos/kv_lens increments) in ONE launch.  Replaces torch's
// reduce_kernel argmax (measured 43 us for [1,128k] bf16 — a 6 GB/s
// config) + index_copy + copy_ + three adds = ~6 dispatches per decode
// step (profiles/r02_fp8_decode_kernel_stats.txt).  Tie-break matches
// torch.argmax: lowest index wins.  The ctr increment stays a separate
// captured op so every block reads the same ring row (stream order
// makes that race-free inside the graph).
extern "C" __global__ __launch_bounds__(256) void decode_advance_kernel(
    const unsigned short* __restrict__ logits,  // [b, V] bf16
    long long V, int* __restrict__ stage,       // [6, b] int32
    long long b, long long* __restrict__ ring,  // [chunk, b] int64
    const long long* __restrict__ ctr,          // [1]
    int chunk) {
  const int i = blockIdx.x;  // batch row
  const unsigned short* row = logits + (long long)i * V;
  float best = -INFINITY;
  int bidx = 0;
  for (long long j = threadIdx.x; j < V; j += 256) {
    float v = bf2f(row[j]);
    if (v > best) {
      best = v;
      bidx = (int)j;
    }
  }
  __shared__ float smax[256];
  __shared__ int sidx[256];
  smax[threadIdx.x] = best;
  sidx[threadIdx.x] = bidx;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if (threadIdx.x < s) {
      float a = smax[threadIdx.x], c = smax[threadIdx.x + s];
      int ai = sidx[threadIdx.x], ci = sidx[threadIdx.x + s];
      if (c > a || (c == a && ci < ai)) {
        smax[threadIdx.x] = c;
        sidx[threadIdx.x] = ci;
      }
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    const int tok = sidx[0];
    ring[(*ctr % chunk) * b + i] = (long long)tok;
    stage[0 * b + i] = tok;   // next token fed back into the graph
    stage[1 * b + i] += 1;    // positions
    stage[3 * b + i] += 1;    // pos
    stage[4 * b + i] += 1;    // kv_lens
  }
}

extern "C" void decode_advance_launch(const void* logits, long long V,
                                      void* stage, long long b, void* ring,
                                      const void* ctr, int chunk,
                                      hipStream_t stream) {
  hipLaunchKernelGGL(decode_advance_kernel, dim3((unsigned)b), dim3(256), 0,
                     stream, (const unsigned short*)logits, V, (int*)stage,
                     b, (long long*)ring, (const long long*)ctr, chunk);
}
