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

  // Online softmax over this wave's KV slice; o accumulator: 2 d per
  // lane.  TWO independent chains per wave (even/odd rows of the
  // slice, merged below): the per-row dot -> wave-reduce -> exp ->
  // rescale dependency chain is the latency bound at long kv, and the
  // second chain fills its stalls.
  float m_c[2] = {-INFINITY, -INFINITY};
  float l_c[2] = {0.f, 0.f}, o0_c[2] = {0.f, 0.f}, o1_c[2] = {0.f, 0.f};
  int s = w;
  for (; s + 4 < kv_len; s += 8) {
    const unsigned short* krow0 = Kb + s * kv_rowstride;
    const unsigned short* krow1 = Kb + (s + 4) * kv_rowstride;
    float d0 = q0 * bf2f(krow0[lane]) + q1 * bf2f(krow0[lane + 64]);
    float d1 = q0 * bf2f(krow1[lane]) + q1 * bf2f(krow1[lane + 64]);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      d0 += __shfl_xor(d0, off, 64);
      d1 += __shfl_xor(d1, off, 64);
    }
    const unsigned short* vrow0 = Vb + s * kv_rowstride;
    const unsigned short* vrow1 = Vb + (s + 4) * kv_rowstride;
    float mn0 = fmaxf(m_c[0], d0);
    float mn1 = fmaxf(m_c[1], d1);
    float c0 = (m_c[0] == -INFINITY) ? 0.f : __expf(m_c[0] - mn0);
    float c1 = (m_c[1] == -INFINITY) ? 0.f : __expf(m_c[1] - mn1);
    float e0 = __expf(d0 - mn0);
    float e1 = __expf(d1 - mn1);
    o0_c[0] = o0_c[0] * c0 + e0 * bf2f(vrow0[lane]);
    o1_c[0] = o1_c[0] * c0 + e0 * bf2f(vrow0[lane + 64]);
    o0_c[1] = o0_c[1] * c1 + e1 * bf2f(vrow1[lane]);
    o1_c[1] = o1_c[1] * c1 + e1 * bf2f(vrow1[lane + 64]);
    l_c[0] = l_c[0] * c0 + e0;
    l_c[1] = l_c[1] * c1 + e1;
    m_c[0] = mn0;
    m_c[1] = mn1;
  }
  for (; s < kv_len; s += 4) {
    const unsigned short* krow = Kb + s * kv_rowstride;
    float dot = q0 * bf2f(krow[lane]) + q1 * bf2f(krow[lane + 64]);
    dot = wave_reduce_sum(dot);
    float m_new = fmaxf(m_c[0], dot);
    float corr = (m_c[0] == -INFINITY) ? 0.f : __expf(m_c[0] - m_new);
    float e = __expf(dot - m_new);
    const unsigned short* vrow = Vb + s * kv_rowstride;
    o0_c[0] = o0_c[0] * corr + e * bf2f(vrow[lane]);
    o1_c[0] = o1_c[0] * corr + e * bf2f(vrow[lane + 64]);
    l_c[0] = l_c[0] * corr + e;
    m_c[0] = m_new;
  }
  // merge the two chains
  float m = fmaxf(m_c[0], m_c[1]);
  float cc0 = (m_c[0] == -INFINITY) ? 0.f : __expf(m_c[0] - m);
  float cc1 = (m_c[1] == -INFINITY) ? 0.f : __expf(m_c[1] - m);
  float l = l_c[0] * cc0 + l_c[1] * cc1;
  float o0 = o0_c[0] * cc0 + o0_c[1] * cc1;
  float o1 = o1_c[0] * cc0 + o1_c[1] * cc1;
  if (m == -INFINITY) {
    l = 0.f;
    o0 = o1 = 0.f;
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
