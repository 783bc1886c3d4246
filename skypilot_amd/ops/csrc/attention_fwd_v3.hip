// Flash-attention forward v3 — 8-wave 32x32 swapped-operand schedule.
//
// The round-2 deep-pipeline forward (guide §B "8-warp 32x32 ladder",
// techniques T2/T5/T10/T12/T13/T14 composed, not grafted):
//   - 512-thread blocks: 8 waves, wave w owns q rows [32w, 32w+32) of a
//     256-row q block; kv tiles of 64.
//   - Swapped QK^T (S^T = K Q^T): each lane holds 32 kv-scores of ONE q
//     column in MFMA C-registers, so the online softmax is 31 in-lane
//     fmax + one lane<->lane+32 exchange — zero LDS softmax state.
//   - K in XOR-swizzled LDS (16-slot swizzle keyed on row&15, rows are
//     256 B), staged by global_load_lds with pre-swizzled SOURCE
//     addresses (linear LDS dest, guide m201 stage pattern).
//   - V in a [kv/4][d/16][4][16] subtiled LDS layout, consumed as the
//     PV A-operand by paired ds_read_b64_tr_b16 (hardware transpose
//     read, guide T10) — no transposed scatter staging.
//   - P^T -> B-fragments by v_cvt_pk_bf16_f32 + permlane32_swap (T12,
//     semantics HW-verified: profiles/r01_hw_probe_semantics.txt).
//   - defer-max rescale threshold (T13, THR=8 in exp2 domain).
//   - One barrier per kv tile: V commits from prefetch registers, then
//     barrier (the compiler's vmcnt drain covers the async K loads),
//     then next tile's loads issue and stay in flight under compute.
//   - Q is pre-scaled by scale*log2(e) so softmax uses native exp2.
//
// Layouts: Q,O [B,S,Hq,128]; K,V [B,S,Hkv,128]; lse [B,Hq,S] fp32
// (natural-log convention, same as v1/v2 kernels).
#include <cstdlib>
#include "common.h"

#define ATT_D 128
#define KVB 64
#define QBW 32
#define NWV3 8
#define QBLK3 (QBW * NWV3)  // 256 q rows per block (8-wave variant)

#include "attn_v3.h"

// NW = waves per block (8 -> one 512-thread block/CU; 4 -> two
// independent 256-thread blocks/CU whose phases interleave freely —
// round-1 measured that cross-block overlap beats one barrier-synced
// big block on these structures).
template <int NW>
__global__ __launch_bounds__(64 * NW, 2) void attn_fwd_v3_t(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, unsigned short* __restrict__ O,
    float* __restrict__ lse_out, int B, int S, int Hq, int Hkv, float scale,
    int causal) {
  constexpr int QB = QBW * NW;        // q rows per block
  constexpr int NKI = KVB / NW / 4;   // K global_load_lds per wave
  constexpr int NVC = 16 / NW;        // V 16B chunks per thread
  __shared__ unsigned short k_lds[2][KVB * ATT_D];  // 2 x 16 KB
  __shared__ unsigned short v_lds[2][KVB * ATT_D];  // 2 x 16 KB

  // Causal grids are PAIRED: one block owns a (heavy, light) q-tile
  // pair — qtA = nq-1-bx (deep kv sweep) then qtB = bx — with the KV
  // staging pipeline running straight through the boundary.  This
  // amortizes the per-q-tile pipeline ramp (measured: causal
  // efficiency rises with tiles/block; mask & wave-skip are null) and
  // load-balances (every pair sweeps nq+1 kv tiles).
  const int nq = S / QB;
  const bool paired = causal && (int)gridDim.x * 2 == nq;
  const int qtA = paired ? nq - 1 - (int)blockIdx.x
                         : (int)gridDim.x - 1 - (int)blockIdx.x;
  const int qtB = paired ? (int)blockIdx.x : 0;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int qh = bh % Hq;
  const int kvh = qh / (Hq / Hkv);
  int qbase = qtA * QB;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int col = lane & 31;  // q column owned by this lane
  const int h = lane >> 5;
  const int g = lane >> 4;
  const int lw = lane & 15;

  const long long q_rowstride = (long long)Hq * ATT_D;
  const long long kv_rowstride = (long long)Hkv * ATT_D;
  const unsigned short* Qb = Q + ((long long)b * S * Hq + qh) * ATT_D;
  const unsigned short* Kb = K + ((long long)b * S * Hkv + kvh) * ATT_D;
  const unsigned short* Vb = V + ((long long)b * S * Hkv + kvh) * ATT_D;

  int my_q = qbase + QBW * w + col;

  // ---- Q as B-fragments, pre-scaled by scale*log2(e) (exp2 softmax).
  // (qs/src recomputed inside the macro from kernel args so nothing
  // stays live across the kv loop for the boundary reload.)
  s16x8 q_b[8];
#define V3_LOAD_Q()                                                       \
  {                                                                       \
    const float qs = scale * 1.44269504088896340736f;                     \
    const unsigned short* src =                                           \
        Q + ((long long)b * S * Hq + qh + (long long)my_q * Hq) * ATT_D;  \
    _Pragma("unroll") for (int ks = 0; ks < 8; ++ks) {                    \
      s16x8 raw = *(const s16x8*)(src + ks * 16 + h * 8);                 \
      _Pragma("unroll") for (int j = 0; j < 8; ++j)                       \
        raw[j] = (short)f2bf(bf2f((unsigned short)raw[j]) * qs);          \
      q_b[ks] = raw;                                                      \
    }                                                                     \
  }
  V3_LOAD_Q();

  f32x16 o_t[4];
#pragma unroll
  for (int dn = 0; dn < 4; ++dn)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_t[dn][r] = 0.f;
  float m_run = -INFINITY, l_run = 0.f;

  // causal: 1 = normal; 2 = debug timing (causal loop structure, mask
  // skipped); 3 = debug timing (no per-wave diagonal skip).
  const int nt_A = causal ? (qbase + QB) / KVB : S / KVB;
  const int nt_B = paired ? (qtB * QB + QB) / KVB : 0;
  const int total_t = nt_A + nt_B;
  // Tiles this wave actually computes (beyond its diagonal: staging +
  // barriers only).
  int w_tiles = (causal == 1 || causal == 2)
                    ? ((qbase + QBW * w + QBW - 1) >> 6) + 1
                    : nt_A;

  // ---- epilogue: O[q][d] = O^T[d][q] / l; lse = ln2*m' + ln(l).
  unsigned short* Ob = O + ((long long)b * S * Hq + qh) * ATT_D;
  float* lse_b = lse_out + ((long long)b * Hq + qh) * S;
#define V3_EPILOGUE()                                                     \
  {                                                                       \
    const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;                \
    unsigned short* orow = Ob + (long long)my_q * q_rowstride;            \
    _Pragma("unroll") for (int dn = 0; dn < 4; ++dn)                      \
      _Pragma("unroll") for (int rq = 0; rq < 4; ++rq) {                  \
        s16x4 ov;                                                         \
        _Pragma("unroll") for (int r = 0; r < 4; ++r)                     \
          ov[r] = (short)f2bf(o_t[dn][rq * 4 + r] * inv_l);               \
        *(s16x4*)(orow + dn * 32 + rq * 8 + h * 4) = ov;                  \
      }                                                                   \
    if (h == 0)                                                           \
      lse_b[my_q] = (l_run > 0.f)                                         \
                        ? 0.69314718055994530942f * m_run +               \
                              __logf(l_run)                               \
                        : -INFINITY;                                      \
  }

  // ---- staging: BOTH tiles by global_load_lds (zero staging
  // registers).  K rows pre-swizzled for the b128 A-fragment reads; V
  // rows linear (tr reads tolerate the resulting ~4-way conflicts;
  // register staging for V's subtiled layout cost 16 VGPRs and
  // spilled the paired kernel).
#define V3_STAGE(kvoff, bufi)                                             \
  {                                                                       \
    _Pragma("unroll") for (int i = 0; i < NKI; ++i) {                     \
      int row = (KVB / NW) * w + 4 * i + (lane >> 4);                     \
      int chunk = lane & 15;                                              \
      long long srow = (kvoff + row) * kv_rowstride;                      \
      gload_lds16((const char*)(Kb + srow) +                              \
                      ((chunk ^ (row & 15)) << 4),                        \
                  (char*)k_lds[bufi] + ((KVB / NW) * w + 4 * i) * 256);   \
      gload_lds16((const char*)(Vb + srow) + (chunk << 4),                \
                  (char*)v_lds[bufi] + ((KVB / NW) * w + 4 * i) * 256);   \
    }                                                                     \
  }
  V3_STAGE((long long)0, 0);

  for (int it = 0; it < total_t; ++it) {
    if (paired && it == nt_A) {
      // ---- boundary: finish the heavy tile, switch to the light one.
      V3_EPILOGUE();
      qbase = qtB * QB;
      my_q = qbase + QBW * w + col;
      V3_LOAD_Q();
#pragma unroll
      for (int dn = 0; dn < 4; ++dn)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_t[dn][r] = 0.f;
      m_run = -INFINITY;
      l_run = 0.f;
      w_tiles = (causal == 3) ? nt_B
                              : ((qbase + QBW * w + QBW - 1) >> 6) + 1;
    }
    const int kt = it < nt_A ? it : it - nt_A;
    const int buf = it & 1;
    // one barrier per tile; the explicit vm_drain is REQUIRED: hipcc's
    // pre-barrier waitcnt is lgkmcnt(0) only and does NOT cover the
    // in-flight global_load_lds for this buffer (see attn_v3.h).
    vm_drain();
    __syncthreads();
    // issue next tile's staging (possibly the light q-tile's tile 0:
    // the pipeline never drains at the boundary).
    if (it + 1 < total_t) {
      const int kt2 = (it + 1 < nt_A) ? it + 1 : it + 1 - nt_A;
      V3_STAGE((long long)kt2 * KVB, buf ^ 1);
    }
    if (kt >= w_tiles) continue;  // past this wave's diagonal

    const int kvbase = kt * KVB;
    // ---- S^T = K Q^T (Q pre-scaled).  2 kv blocks x 8 k-steps,
    // kv-inner so two accumulator chains interleave.
    f32x16 st[2];
#pragma unroll
    for (int n = 0; n < 2; ++n)
#pragma unroll
      for (int r = 0; r < 16; ++r) st[n][r] = 0.f;
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 8; ++ks)
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        int krow = n * 32 + col;
        s16x8 kf = *(const s16x8*)(
            (char*)k_lds[buf] + swzK16(krow * 256 + (ks * 2 + h) * 16, krow));
        st[n] = MFMA32V3(as_bf16x8(kf), as_bf16x8(q_b[ks]), st[n]);
      }
    __builtin_amdgcn_s_setprio(0);

    // ---- causal mask (finite big-negative so exp2 underflows to 0;
    // m_run is already real for every row because tile 0 is unmasked).
    if (causal == 1 && kvbase + KVB - 1 > qbase + QBW * w) {
#pragma unroll
      for (int n = 0; n < 2; ++n)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kv = kvbase + n * 32 + (r & 3) + ((r >> 2) << 3) + (h << 2);
          if (kv > my_q) st[n][r] = -30000.f;
        }
    }

    // ---- issue the first PV tr-read batch NOW: its ~latency hides
    // under the (LGKM-free) softmax VALU work below.
    const int kv0g = (g >> 1) << 3;
    const int trd = ((g & 1) << 4) + ((lw & 3) << 2);
    const int trr = lw >> 2;
    const unsigned vbase = (unsigned)(size_t)((char*)v_lds[buf]);
    tr4 tA0, tB0;  // tr-read staging (issue-early, pack under latency)
#define V3_ISSUE(tA, tB, ks)                                              \
  {                                                                       \
    int kvr = (ks) * 16 + kv0g + trr;                                     \
    ds_tr4_issue(&tA, vbase + kvr * 256 + 2 * trd,                        \
                 vbase + (kvr + 4) * 256 + 2 * trd,                       \
                 vbase + kvr * 256 + 2 * (32 + trd),                      \
                 vbase + (kvr + 4) * 256 + 2 * (32 + trd));               \
    ds_tr4_issue(&tB, vbase + kvr * 256 + 2 * (64 + trd),                 \
                 vbase + (kvr + 4) * 256 + 2 * (64 + trd),                \
                 vbase + kvr * 256 + 2 * (96 + trd),                      \
                 vbase + (kvr + 4) * 256 + 2 * (96 + trd));               \
  }
    V3_ISSUE(tA0, tB0, 0);

    // ---- online softmax, latency-shaped:
    //   - max is a TREE reduction (a linear fmax chain is 32 dependent
    //     VALU ops ~128 cyc; the tree is depth 5),
    //   - cross-half exchanges use permlane (VALU), not __shfl_xor's
    //     ds_bpermute, so no compiler lgkmcnt(0) drains in-flight tr
    //     reads,
    //   - exp + sum are SLICED into the PV loop (8 values per k-step,
    //     exactly the slice each P-pack consumes) where they hide under
    //     the MFMA clusters (m214 "sm-split").
    float tmax[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) tmax[r] = fmaxf(st[0][r], st[1][r]);
#pragma unroll
    for (int s = 8; s > 0; s >>= 1)
#pragma unroll
      for (int r = 0; r < s; ++r) tmax[r] = fmaxf(tmax[r], tmax[r + s]);
    float pmax = fmaxf(tmax[0], xhalf32(tmax[0]));
    // defer-max (T13): skip the O-rescale while the running max grows
    // by <= 8 (P bounded by 2^8; f32 accumulators absorb it).
    if (!__all(pmax - m_run <= 8.0f)) {
      float m_new = fmaxf(m_run, pmax);
      float corr = (m_run == -INFINITY) ? 0.f : exp2f(m_run - m_new);
#pragma unroll
      for (int dn = 0; dn < 4; ++dn)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_t[dn][r] *= corr;
      l_run *= corr;
      m_run = m_new;
    }

    // exp + sum (pairwise sums keep the dependence shallow).
    float rs = 0.f;
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      float part = 0.f;
#pragma unroll
      for (int r = 0; r < 16; r += 2) {
        float e0 = exp2f(st[n][r] - m_run);
        float e1 = exp2f(st[n][r + 1] - m_run);
        st[n][r] = e0;
        st[n][r + 1] = e1;
        part += e0 + e1;
      }
      rs += part;
    }
    rs += xhalf32(rs);
    l_run += rs;

    // ---- PV, 2-deep pipelined: batch ks+1's 8 tr reads are issued
    // before batch ks's counted wait (lgkmcnt(8) leaves them in
    // flight); P-pack runs under tr latency.
#define V3_MFMA(tA, tB, pb)                                               \
  {                                                                       \
    union { unsigned long long u[2]; s16x8 v; } vf;                       \
    __builtin_amdgcn_s_setprio(1);                                        \
    vf.u[0] = tA.d[0];                                                    \
    vf.u[1] = tA.d[1];                                                    \
    o_t[0] = MFMA32V3(as_bf16x8(vf.v), as_bf16x8(pb), o_t[0]);            \
    vf.u[0] = tA.d[2];                                                    \
    vf.u[1] = tA.d[3];                                                    \
    o_t[1] = MFMA32V3(as_bf16x8(vf.v), as_bf16x8(pb), o_t[1]);            \
    vf.u[0] = tB.d[0];                                                    \
    vf.u[1] = tB.d[1];                                                    \
    o_t[2] = MFMA32V3(as_bf16x8(vf.v), as_bf16x8(pb), o_t[2]);            \
    vf.u[0] = tB.d[2];                                                    \
    vf.u[1] = tB.d[3];                                                    \
    o_t[3] = MFMA32V3(as_bf16x8(vf.v), as_bf16x8(pb), o_t[3]);            \
    __builtin_amdgcn_s_setprio(0);                                        \
  }
    s16x8 pb0;
    V3_PACK2(pb0, st, 0);
    lgkm_wait0_bind2(&tA0, &tB0);
    V3_MFMA(tA0, tB0, pb0);
    V3_ISSUE(tA0, tB0, 1);
    V3_PACK2(pb0, st, 1);
    lgkm_wait0_bind2(&tA0, &tB0);
    V3_MFMA(tA0, tB0, pb0);
    V3_ISSUE(tA0, tB0, 2);
    V3_PACK2(pb0, st, 2);
    lgkm_wait0_bind2(&tA0, &tB0);
    V3_MFMA(tA0, tB0, pb0);
    V3_ISSUE(tA0, tB0, 3);
    V3_PACK2(pb0, st, 3);
    lgkm_wait0_bind2(&tA0, &tB0);
    V3_MFMA(tA0, tB0, pb0);
  }

  V3_EPILOGUE();
#undef V3_EPILOGUE
#undef V3_LOAD_Q
#undef V3_ISSUE
}

extern "C" void attn_fwd_v3_launch(const void* Q, const void* K,
                                   const void* V, void* O, float* lse, int B,
                                   int S, int Hq, int Hkv, float scale,
                                   bool causal, hipStream_t stream) {
  static const int nw = [] {
    const char* e = getenv("SKY_ATTN_FWD_V3_NW");
    return e ? atoi(e) : 4;
  }();
  static const int dbg = [] {
    const char* e = getenv("SKY_ATTN_DEBUG_MODE");
    return e ? atoi(e) : 0;
  }();
  const int cz = causal ? (dbg ? dbg : 1) : 0;
  static const int pair = [] {
    const char* e = getenv("SKY_ATTN_FWD_V3_PAIR");
    return e ? atoi(e) : 1;
  }();
  if (nw == 8) {
    int nq = S / (QBW * 8);
    int gx = (pair && cz && nq % 2 == 0) ? nq / 2 : nq;
    dim3 grid(gx, B * Hq);
    hipLaunchKernelGGL(attn_fwd_v3_t<8>, grid, dim3(512), 0, stream,
                       (const unsigned short*)Q, (const unsigned short*)K,
                       (const unsigned short*)V, (unsigned short*)O, lse, B,
                       S, Hq, Hkv, scale, cz);
    return;
  }
  int nq = S / (QBW * 4);
  int gx = (pair && cz && nq % 2 == 0) ? nq / 2 : nq;
  dim3 grid(gx, B * Hq);
  hipLaunchKernelGGL(attn_fwd_v3_t<4>, grid, dim3(256), 0, stream,
                     (const unsigned short*)Q, (const unsigned short*)K,
                     (const unsigned short*)V, (unsigned short*)O, lse, B, S,
                     Hq, Hkv, scale, cz);
}
