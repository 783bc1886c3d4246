// Decode attention (single query token vs KV cache) for CDNA4/gfx950.
//
// Memory-bound KV read (guide App. B "Attention decode"): one 256-thread
// block per (sequence, q-head); the 4 waves split the KV length, each
// doing an online-softmax partial with vectorized s16x8 K/V loads; the
// partials merge through LDS.  Supports GQA and ragged batches
// (per-sequence kv_len) over a [B, S_max, Hkv, D] cache.
#include "common.h"

#define DEC_D 128

extern "C" __global__ __launch_bounds__(256) void attn_decode_kernel(
    const unsigned short* __restrict__ Q,    // [B, Hq, D]
    const unsigned short* __restrict__ Kc,   // [B, S_max, Hkv, D]
    const unsigned short* __restrict__ Vc,
    unsigned short* __restrict__ O,          // [B, Hq, D]
    const int* __restrict__ kv_lens,         // [B] (indexed by Q batch)
    const int* __restrict__ slot_ids,        // [B] -> cache slot
    int B, int S_max, int Hq, int Hkv, float scale) {
  const int bq = blockIdx.x / Hq;
  const int b = slot_ids ? slot_ids[bq] : bq;
  const int qh = blockIdx.x % Hq;
  const int kvh = qh / (Hq / Hkv);
  const int kv_len = kv_lens[bq];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;

  // Q for this head: 128 bf16 -> each lane holds 2 (lane, lane+64).
  const unsigned short* qv = Q + ((long long)bq * Hq + qh) * DEC_D;
  const float q0 = bf2f(qv[lane]) * scale;
  const float q1 = bf2f(qv[lane + 64]) * scale;

  const long long kv_rowstride = (long long)Hkv * DEC_D;
  const unsigned short* Kb = Kc + ((long long)b * S_max * Hkv + kvh) * DEC_D;
  const unsigned short* Vb = Vc + ((long long)b * S_max * Hkv + kvh) * DEC_D;

  // Online softmax over this wave's KV slice; o accumulator: 2 d per lane.
  float m = -INFINITY, l = 0.f, o0 = 0.f, o1 = 0.f;
  for (int s = w; s < kv_len; s += 4) {
    const unsigned short* krow = Kb + s * kv_rowstride;
    // dot(q, k): each lane contributes 2 elements, wave-reduce.
    float dot = q0 * bf2f(krow[lane]) + q1 * bf2f(krow[lane + 64]);
    dot = wave_reduce_sum(dot);
    float m_new = fmaxf(m, dot);
    float corr = (m == -INFINITY) ? 0.f : __expf(m - m_new);
    float e = __expf(dot - m_new);
    const unsigned short* vrow = Vb + s * kv_rowstride;
    o0 = o0 * corr + e * bf2f(vrow[lane]);
    o1 = o1 * corr + e * bf2f(vrow[lane + 64]);
    l = l * corr + e;
    m = m_new;
  }

  // Merge the 4 wave partials via LDS.
  __shared__ float red_m[4], red_l[4];
  __shared__ float red_o[4][DEC_D];
  if (lane == 0) {
    red_m[w] = m;
    red_l[w] = l;
  }
  red_o[w][lane] = o0;
  red_o[w][lane + 64] = o1;
  __syncthreads();
  if (w == 0) {
    float gm = -INFINITY;
#pragma unroll
    for (int i = 0; i < 4; ++i) gm = fmaxf(gm, red_m[i]);
    float gl = 0.f, a0 = 0.f, a1 = 0.f;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      float c = (red_m[i] == -INFINITY) ? 0.f : __expf(red_m[i] - gm);
      gl += red_l[i] * c;
      a0 += red_o[i][lane] * c;
      a1 += red_o[i][lane + 64] * c;
    }
    float inv = (gl > 0.f) ? 1.f / gl : 0.f;
    unsigned short* orow = O + ((long long)bq * Hq + qh) * DEC_D;
    orow[lane] = f2bf(a0 * inv);
    orow[lane + 64] = f2bf(a1 * inv);
  }
}

extern "C" void attn_decode_launch(const void* Q, const void* Kc,
                                   const void* Vc, void* O,
                                   const int* kv_lens, const int* slot_ids,
                                   int B, int S_max, int Hq, int Hkv,
                                   float scale, hipStream_t stream) {
  hipLaunchKernelGGL(attn_decode_kernel, dim3(B * Hq), dim3(256), 0, stream,
                     (const unsigned short*)Q, (const unsigned short*)Kc,
                     (const unsigned short*)Vc, (unsigned short*)O, kv_lens,
                     slot_ids, B, S_max, Hq, Hkv, scale);
}
