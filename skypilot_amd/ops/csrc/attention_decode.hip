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
#define DEC_CHAINS 4
  float m_c[DEC_CHAINS], l_c[DEC_CHAINS];
  float o0_c[DEC_CHAINS], o1_c[DEC_CHAINS];
#pragma unroll
  for (int c = 0; c < DEC_CHAINS; ++c) {
    m_c[c] = -INFINITY;
    l_c[c] = 0.f;
    o0_c[c] = 0.f;
    o1_c[c] = 0.f;
  }
  int s = w;
  for (; s + 4 * (DEC_CHAINS - 1) < kv_len; s += 4 * DEC_CHAINS) {
    float d[DEC_CHAINS];
#pragma unroll
    for (int c = 0; c < DEC_CHAINS; ++c) {
      const unsigned short* krow = Kb + (s + 4 * c) * kv_rowstride;
      d[c] = q0 * bf2f(krow[lane]) + q1 * bf2f(krow[lane + 64]);
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
#pragma unroll
      for (int c = 0; c < DEC_CHAINS; ++c)
        d[c] += __shfl_xor(d[c], off, 64);
#pragma unroll
    for (int c = 0; c < DEC_CHAINS; ++c) {
      const unsigned short* vrow = Vb + (s + 4 * c) * kv_rowstride;
      float mn = fmaxf(m_c[c], d[c]);
      float cr = (m_c[c] == -INFINITY) ? 0.f : __expf(m_c[c] - mn);
      float e = __expf(d[c] - mn);
      o0_c[c] = o0_c[c] * cr + e * bf2f(vrow[lane]);
      o1_c[c] = o1_c[c] * cr + e * bf2f(vrow[lane + 64]);
      l_c[c] = l_c[c] * cr + e;
      m_c[c] = mn;
    }
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
  // merge the chains
  float m = -INFINITY;
#pragma unroll
  for (int c = 0; c < DEC_CHAINS; ++c) m = fmaxf(m, m_c[c]);
  float l = 0.f, o0 = 0.f, o1 = 0.f;
#pragma unroll
  for (int c = 0; c < DEC_CHAINS; ++c) {
    float cc = (m_c[c] == -INFINITY) ? 0.f : __expf(m_c[c] - m);
    l += l_c[c] * cc;
    o0 += o0_c[c] * cc;
    o1 += o1_c[c] * cc;
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
