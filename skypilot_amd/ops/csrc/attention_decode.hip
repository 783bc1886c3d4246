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

// Fused rope + cache-write + decode attention: consumes the RAW packed
// qkv GEMV output directly.  Removes the separate rope_kvwrite launch
// (32/token, 4.9 us each = ~6% of single-stream decode,
// profiles/r02_fp8_decode_kernel_stats.txt) and the q_out roundtrip.
// Each (seq, q-head) block ropes its own q AND its kv-head's k in
// registers; the current token's k/v never goes through the cache on
// this step (it is attended from registers), so there is no
// write->read ordering hazard: one designated block per kv head
// (qh % group == 0) writes the cache row for FUTURE steps.
extern "C" __global__ __launch_bounds__(256) void attn_decode_qkv_kernel(
    const unsigned short* __restrict__ qkv,  // [B, (Hq+2*Hkv)*D] raw
    unsigned short* __restrict__ Kc,         // [slots, S_max, Hkv, D]
    unsigned short* __restrict__ Vc,
    unsigned short* __restrict__ O,          // [B, Hq, D]
    const float* __restrict__ cos_tab,       // [S, D/2]
    const float* __restrict__ sin_tab,
    const int* __restrict__ positions,       // [B]
    const int* __restrict__ kv_lens,         // [B] == positions+1
    const int* __restrict__ slot_ids,        // [B] -> cache slot
    int B, int S_max, int Hq, int Hkv, float scale) {
  const int bq = blockIdx.x / Hq;
  const int b = slot_ids ? slot_ids[bq] : bq;
  const int qh = blockIdx.x % Hq;
  const int group = Hq / Hkv;
  const int kvh = qh / group;
  const int kv_len = kv_lens[bq];
  const int p = positions[bq];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;

  // rope pair for this lane: (d=lane, d=lane+64), table row p.
  const float cr_ = cos_tab[(long long)p * (DEC_D / 2) + lane];
  const float sr_ = sin_tab[(long long)p * (DEC_D / 2) + lane];

  const int heads = Hq + 2 * Hkv;
  const unsigned short* qraw =
      qkv + ((long long)bq * heads + qh) * DEC_D;
  const unsigned short* kraw =
      qkv + ((long long)bq * heads + Hq + kvh) * DEC_D;
  const unsigned short* vraw =
      qkv + ((long long)bq * heads + Hq + Hkv + kvh) * DEC_D;

  // roped q (pre-scaled) + current-token roped k and v in registers.
  float qa = bf2f(qraw[lane]), qb = bf2f(qraw[lane + 64]);
  const float q0 = (qa * cr_ - qb * sr_) * scale;
  const float q1 = (qb * cr_ + qa * sr_) * scale;
  float ka = bf2f(kraw[lane]), kb = bf2f(kraw[lane + 64]);
  const float k0 = ka * cr_ - kb * sr_;
  const float k1 = kb * cr_ + ka * sr_;
  const float v0 = bf2f(vraw[lane]);
  const float v1 = bf2f(vraw[lane + 64]);

  const long long kv_rowstride = (long long)Hkv * DEC_D;
  unsigned short* Kb = Kc + ((long long)b * S_max * Hkv + kvh) * DEC_D;
  unsigned short* Vb = Vc + ((long long)b * S_max * Hkv + kvh) * DEC_D;

  // cache write for future steps: one block per kv head, wave 0.
  if (qh % group == 0 && w == 0) {
    unsigned short* kdst = Kb + (long long)p * kv_rowstride;
    unsigned short* vdst = Vb + (long long)p * kv_rowstride;
    kdst[lane] = f2bf(k0);
    kdst[lane + 64] = f2bf(k1);
    vdst[lane] = f2bf(v0);
    vdst[lane + 64] = f2bf(v1);
  }

  float m_c[DEC_CHAINS], l_c[DEC_CHAINS];
  float o0_c[DEC_CHAINS], o1_c[DEC_CHAINS];
#pragma unroll
  for (int c = 0; c < DEC_CHAINS; ++c) {
    m_c[c] = -INFINITY;
    l_c[c] = 0.f;
    o0_c[c] = 0.f;
    o1_c[c] = 0.f;
  }
  // wave 0, chain 0: seed with the current token (register k/v).
  if (w == 0) {
    float dc = q0 * k0 + q1 * k1;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) dc += __shfl_xor(dc, off, 64);
    m_c[0] = dc;
    l_c[0] = 1.f;
    o0_c[0] = v0;
    o1_c[0] = v1;
  }
  const int n_cache = kv_len - 1;  // rows 0..p-1 come from the cache
  int s = w;
  for (; s + 4 * (DEC_CHAINS - 1) < n_cache; s += 4 * DEC_CHAINS) {
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
  for (; s < n_cache; s += 4) {
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

extern "C" void attn_decode_qkv_launch(
    const void* qkv, void* Kc, void* Vc, void* O, const void* cos_tab,
    const void* sin_tab, const int* positions, const int* kv_lens,
    const int* slot_ids, int B, int S_max, int Hq, int Hkv, float scale,
    hipStream_t stream) {
  hipLaunchKernelGGL(attn_decode_qkv_kernel, dim3(B * Hq), dim3(256), 0,
                     stream, (const unsigned short*)qkv,
                     (unsigned short*)Kc, (unsigned short*)Vc,
                     (unsigned short*)O, (const float*)cos_tab,
                     (const float*)sin_tab, positions, kv_lens, slot_ids,
                     B, S_max, Hq, Hkv, scale);
}
