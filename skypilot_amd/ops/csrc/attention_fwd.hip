// Flash-attention forward (causal, GQA, D=128, bf16) for CDNA4/gfx950.
//
// MI355X-native design (no reference counterpart — SkyPilot ships no
// kernels): 64 q-rows per 256-thread block (4 waves,
// wave w owns q rows [16w,16w+16)), kv tiles of 64, MFMA 16x16x32 bf16.
// K tile and transposed-V tile live in XOR-swizzled LDS so every MFMA
// B-fragment is one conflict-free ds_read_b128; P round-trips through a
// swizzled LDS tile to move from MFMA C-layout to A-layout.  Online
// softmax (running max/sum) is held in registers; log-sum-exp is written
// for the backward pass.
//
// Layouts: Q,O [B,S,Hq,D]; K,V [B,S,Hkv,D]; lse [B,Hq,S] fp32.
#include <cstdlib>
#include "common.h"

#define ATT_D 128
#define BM 64  // q rows per block
#define BN 64  // kv rows per tile
#define NW 4   // waves per block

extern "C" __global__ __launch_bounds__(256, 2) void attn_fwd_kernel(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, unsigned short* __restrict__ O,
    float* __restrict__ lse_out, int B, int S, int Hq, int Hkv, float scale,
    int causal) {
  __shared__ unsigned short k_lds[BN * ATT_D];    // [kv][d], swizzled, 16KB
  __shared__ unsigned short vt_lds[ATT_D * BN];   // [d][kv], swizzled, 16KB
  __shared__ unsigned short p_lds[BM * BN];       // [q][kv], swizzled, 8KB

  // Heavy blocks (large qt => many causal kv tiles) launch first so the
  // dispatch tail is short blocks (load balance across 256 CUs).
  const int qt = gridDim.x - 1 - blockIdx.x;  // q tile index
  const int bh = blockIdx.y;            // b * Hq + qh
  const int b = bh / Hq;
  const int qh = bh % Hq;
  const int kvh = qh / (Hq / Hkv);
  const int qbase = qt * BM;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;               // wave id: q rows [16w, 16w+16)
  const int lrow = lane & 15;           // fragment row/col index
  const int lgrp = lane >> 4;           // fragment k-group

  const long long q_rowstride = (long long)Hq * ATT_D;
  const long long kv_rowstride = (long long)Hkv * ATT_D;
  const unsigned short* Qb = Q + ((long long)b * S * Hq + qh) * ATT_D;
  const unsigned short* Kb = K + ((long long)b * S * Hkv + kvh) * ATT_D;
  const unsigned short* Vb = V + ((long long)b * S * Hkv + kvh) * ATT_D;

  // ---- Q fragments for this wave: rows qbase+16w+lrow, 4 k-subtiles.
  s16x8 a_q[4];
  {
    const int qrow = qbase + 16 * w + lrow;
    const unsigned short* src = Qb + (long long)qrow * q_rowstride;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks)
      a_q[ks] = *(const s16x8*)(src + ks * 32 + lgrp * 8);
  }

  // ---- online-softmax state: 4 rows per lane (rows lgrp*4 + r).
  float m_run[4], l_run[4];
  f32x4 o_acc[8];  // 8 d col-tiles
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -INFINITY;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) o_acc[ct] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int n_kv_tiles = causal ? (qbase + BM + BN - 1) / BN : (S + BN - 1) / BN;

  // Async-STAGE split (guide §6 G15 / T14): each tile's K/V global loads
  // are issued one iteration early into registers, hiding HBM latency
  // under the previous tile's QK^T + softmax; the LDS writes happen at
  // the top of the owning iteration.  Thread t owns chunks
  // (row t/16 + 16i, 16B-chunk t%16), i = 0..3.
  // Staging map: chunk = (tid&7)|((i&1)<<3), row = (tid>>3)|((i>>1)<<5).
  // A wave then covers 8 rows x 8 chunks per instruction: the transposed
  // scatter writes spread over 4(row-words) x 8(d-groups) = 32 banks
  // (the old 4-row map peaked at 16 banks = 4-way conflicts), while the
  // global reads stay 128B-segment coalesced.
  s16x8 kpre[4], vpre[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int s_ch = (tid & 7) | ((i & 1) << 3);
    int s_row = (tid >> 3) | ((i >> 1) << 5);
    long long r = (long long)s_row * kv_rowstride + s_ch * 8;
    kpre[i] = *(const s16x8*)(Kb + r);
    vpre[i] = *(const s16x8*)(Vb + r);
  }

  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    // ---- stage K tile [kv][d] (swizzled) + V tile transposed [d][kv]
    // from the prefetch registers.
    __syncthreads();
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int s_ch = (tid & 7) | ((i & 1) << 3);
      int s_row = (tid >> 3) | ((i >> 1) << 5);
      *(s16x8*)((char*)k_lds + swz(s_row * 256 + s_ch * 16, s_row)) =
          kpre[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int d = s_ch * 8 + j;
        *(unsigned short*)((char*)vt_lds + swzT(d * 128 + s_row * 2, d)) =
            (unsigned short)vpre[i][j];
      }
    }
    __syncthreads();
    if (kt + 1 < n_kv_tiles) {
      // Issue next tile's loads now; they stay in flight through QK^T +
      // softmax (the sync before PV drains vmcnt).
      const long long base = (long long)(kt + 1) * BN * kv_rowstride;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int s_ch = (tid & 7) | ((i & 1) << 3);
        int s_row = (tid >> 3) | ((i >> 1) << 5);
        long long r = base + (long long)s_row * kv_rowstride + s_ch * 8;
        kpre[i] = *(const s16x8*)(Kb + r);
        vpre[i] = *(const s16x8*)(Vb + r);
      }
    }
    const int kvbase = kt * BN;

    // ---- S = scale * Q K^T for this wave's 16 rows x 64 cols.
    f32x4 s_acc[4];
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
      s_acc[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        int krow = ct * 16 + lrow;
        s16x8 bfrag = *(const s16x8*)((char*)k_lds +
                                      swz(krow * 256 + (ks * 32 + lgrp * 8) * 2, krow));
        s_acc[ct] = MFMA_BF16(as_bf16x8(a_q[ks]), as_bf16x8(bfrag), s_acc[ct]);
      }
    }

    // ---- mask + online softmax.  P is streamed straight to LDS (no
    // p[4][4] register tile — saves 16 VGPRs of occupancy).
    const int my_qrow = qbase + 16 * w + lgrp * 4;  // + r
    // Interior tiles (fully below the diagonal, fully in range) skip the
    // per-element mask — wave-uniform branch, most tiles qualify.
    const bool need_mask =
        (causal && kvbase + BN - 1 > qbase + 16 * w) || (kvbase + BN > S);
    float corr[4], m_safe[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = -INFINITY;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        float s = s_acc[ct][r] * scale;
        if (need_mask) {
          int col = kvbase + ct * 16 + lrow;
          if ((causal && col > my_qrow + r) || col >= S) s = -INFINITY;
        }
        s_acc[ct][r] = s;
        mx = fmaxf(mx, s);
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off, 64));
      float m_new = fmaxf(m_run[r], mx);
      // All -inf row (above diagonal): keep m at -inf, contribute nothing.
      m_safe[r] = (m_new == -INFINITY) ? 0.f : m_new;
      corr[r] = (m_run[r] == -INFINITY) ? 0.f : __expf(m_run[r] - m_safe[r]);
      m_run[r] = m_new;
    }
    // p_lds has been free since the top-of-loop barrier (prior PV read it
    // before that); no extra barrier needed before writing P.
    float rs[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ct = 0; ct < 4; ++ct)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float e = (s_acc[ct][r] == -INFINITY)
                      ? 0.f : __expf(s_acc[ct][r] - m_safe[r]);
        rs[r] += e;
        int prow = 16 * w + lgrp * 4 + r;
        int pcol = ct * 16 + lrow;
        *(unsigned short*)((char*)p_lds + swzP(prow * 128 + pcol * 2, prow)) =
            f2bf_trunc(e);
      }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        rs[r] += __shfl_xor(rs[r], off, 64);
      l_run[r] = l_run[r] * corr[r] + rs[r];
    }
    // rescale existing O.
#pragma unroll
    for (int ct = 0; ct < 8; ++ct)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[ct][r] *= corr[r];
    __syncthreads();

    // ---- O += P V : A from p_lds, B from vt_lds.
#pragma unroll
    for (int ct = 0; ct < 8; ++ct) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        int prow = 16 * w + lrow;
        s16x8 afrag = *(const s16x8*)((char*)p_lds +
                                      swzP(prow * 128 + (ks * 32 + lgrp * 8) * 2, prow));
        int vrow = ct * 16 + lrow;
        s16x8 bfrag = *(const s16x8*)((char*)vt_lds +
                                      swzT(vrow * 128 + (ks * 32 + lgrp * 8) * 2, vrow));
        o_acc[ct] = MFMA_BF16(as_bf16x8(afrag), as_bf16x8(bfrag), o_acc[ct]);
      }
    }
  }

  // ---- epilogue: O /= l, write bf16; lse = m + log(l).
  unsigned short* Ob = O + ((long long)b * S * Hq + qh) * ATT_D;
  float* lse_b = lse_out + ((long long)b * Hq + qh) * S;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = qbase + 16 * w + lgrp * 4 + r;
    float inv_l = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
    unsigned short* orow = Ob + (long long)qrow * q_rowstride;
#pragma unroll
    for (int ct = 0; ct < 8; ++ct)
      orow[ct * 16 + lrow] = f2bf(o_acc[ct][r] * inv_l);
    if (lrow == 0)
      lse_b[qrow] = (l_run[r] > 0.f) ? m_run[r] + __logf(l_run[r]) : -INFINITY;
  }
}


// ---------------------------------------------------------------------------
// 32x32x16 variant: 128 q rows per block (wave owns 32), kv tiles of 64.
// Doubles FLOPs per MFMA and per staged byte vs the 16x16 kernel; used
// whenever S % 128 == 0 (the flagship shapes).  Layouts verified by
// tests/test_gpu_ops.py::test_mfma32_layout_probe.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(16))) float f32x16;
#define MFMA32_BF16(a, b, c) \
  __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0)
#define BM32 128

extern "C" __global__ __launch_bounds__(256, 2) void attn_fwd_kernel32(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, unsigned short* __restrict__ O,
    float* __restrict__ lse_out, int B, int S, int Hq, int Hkv, float scale,
    int causal) {
  __shared__ unsigned short k_lds[BN * ATT_D];     // [kv][d]   16KB
  __shared__ unsigned short vt_lds[ATT_D * BN];    // [d][kv]   16KB
  __shared__ unsigned short p_lds[BM32 * BN];      // [q][kv]   16KB
  __shared__ float m_lds[BM32], l_lds[BM32];

  const int qt = gridDim.x - 1 - blockIdx.x;  // heavy blocks first
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int qh = bh % Hq;
  const int kvh = qh / (Hq / Hkv);
  const int qbase = qt * BM32;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;        // wave owns q rows [32w, 32w+32)
  const int col = lane & 31;
  const int half = lane >> 5;

  const long long q_rowstride = (long long)Hq * ATT_D;
  const long long kv_rowstride = (long long)Hkv * ATT_D;
  const unsigned short* Qb = Q + ((long long)b * S * Hq + qh) * ATT_D;
  const unsigned short* Kb = K + ((long long)b * S * Hkv + kvh) * ATT_D;
  const unsigned short* Vb = V + ((long long)b * S * Hkv + kvh) * ATT_D;

  // Q fragments: rows qbase+32w+col, 8 k-subtiles of 16.
  s16x8 a_q[8];
  {
    const unsigned short* src =
        Qb + (long long)(qbase + 32 * w + col) * q_rowstride;
#pragma unroll
    for (int ks = 0; ks < 8; ++ks)
      a_q[ks] = *(const s16x8*)(src + ks * 16 + half * 8);
  }

  f32x16 o_acc[4];
#pragma unroll
  for (int nt = 0; nt < 4; ++nt)
#pragma unroll
    for (int r = 0; r < 16; ++r) o_acc[nt][r] = 0.f;
  for (int i = tid; i < BM32; i += 256) {
    m_lds[i] = -INFINITY;
    l_lds[i] = 0.f;
  }

  const int n_kv_tiles =
      causal ? (qbase + BM32 + BN - 1) / BN : (S + BN - 1) / BN;

  // No T14 reg-prefetch here: the 32x32 accumulators already fill the
  // register budget (prefetch regs caused scratch spills).
  const int pre_row = tid >> 4, pre_ch = tid & 15;
  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int kvbase = kt * BN;
    __syncthreads();
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = pre_row + 16 * i;
      long long src = (long long)(kvbase + row) * kv_rowstride + pre_ch * 8;
      s16x8 kv8 = *(const s16x8*)(Kb + src);
      *(s16x8*)((char*)k_lds + swz(row * 256 + pre_ch * 16, row)) = kv8;
      s16x8 vv8 = *(const s16x8*)(Vb + src);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int d = pre_ch * 8 + j;
        *(unsigned short*)((char*)vt_lds + swzT(d * 128 + row * 2, d)) =
            (unsigned short)vv8[j];
      }
    }
    __syncthreads();

    // S = scale * Q K^T : wave's 32 rows x 64 cols (2 col-tiles of 32).
    f32x16 s_acc[2];
#pragma unroll
    for (int nt = 0; nt < 2; ++nt)
#pragma unroll
      for (int r = 0; r < 16; ++r) s_acc[nt][r] = 0.f;
    // ks outer / nt inner: interleaves the two accumulator chains so the
    // ~8-cycle MFMA latency is covered by independent issues (nt-outer
    // made every MFMA depend on the previous one: 2x slowdown measured).
#pragma unroll
    for (int ks = 0; ks < 8; ++ks) {
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        int krow = nt * 32 + col;
        s16x8 bfrag = *(const s16x8*)(
            (char*)k_lds + swz(krow * 256 + (ks * 16 + half * 8) * 2, krow));
        s_acc[nt] = MFMA32_BF16(as_bf16x8(a_q[ks]), as_bf16x8(bfrag),
                                s_acc[nt]);
      }
    }

    // Mask + online softmax; P streamed to LDS.
    const bool need_mask =
        (causal && kvbase + BN - 1 > qbase + 32 * w) || (kvbase + BN > S);
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int my_row = qbase + 32 * w + (r & 3) + 8 * (r >> 2) + 4 * half;
      float mx = -INFINITY;
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        float s = s_acc[nt][r] * scale;
        if (need_mask) {
          int c2 = kvbase + nt * 32 + col;
          if ((causal && c2 > my_row) || c2 >= S) s = -INFINITY;
        }
        s_acc[nt][r] = s;
        mx = fmaxf(mx, s);
      }
#pragma unroll
      for (int off = 16; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, 64));
      const int prow = 32 * w + (r & 3) + 8 * (r >> 2) + 4 * half;
      float m_old = m_lds[prow];
      float m_new = fmaxf(m_old, mx);
      float m_safe = (m_new == -INFINITY) ? 0.f : m_new;
      float corr = (m_old == -INFINITY) ? 0.f : __expf(m_old - m_safe);
      if (col == 0) m_lds[prow] = m_new;
      float rs = 0.f;
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        float e = (s_acc[nt][r] == -INFINITY)
                      ? 0.f : __expf(s_acc[nt][r] - m_safe);
        rs += e;
        int pcol = nt * 32 + col;
        *(unsigned short*)((char*)p_lds + swzP(prow * 128 + pcol * 2, prow)) =
            f2bf_trunc(e);
      }
#pragma unroll
      for (int off = 16; off > 0; off >>= 1) rs += __shfl_xor(rs, off, 64);
      if (col == 0) l_lds[prow] = l_lds[prow] * corr + rs;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) o_acc[nt][r] *= corr;
    }
    // (no barrier: p_lds rows 32w..32w+31 are wave-private, and next
    // tile's k/v staging is ordered by the loop-top barrier)

    // O += P V : A-frags from p_lds (4, reused), B from vt_lds.
    s16x8 pa[4];
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      int prow = 32 * w + col;
      pa[ks] = *(const s16x8*)(
          (char*)p_lds + swzP(prow * 128 + (ks * 16 + half * 8) * 2, prow));
    }
#pragma unroll
    for (int ks = 0; ks < 4; ++ks)
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        int vrow = nt * 32 + col;
        s16x8 bfrag = *(const s16x8*)(
            (char*)vt_lds + swzT(vrow * 128 + (ks * 16 + half * 8) * 2,
                                 vrow));
        o_acc[nt] = MFMA32_BF16(as_bf16x8(pa[ks]), as_bf16x8(bfrag),
                                o_acc[nt]);
      }
  }

  // Epilogue.
  unsigned short* Ob = O + ((long long)b * S * Hq + qh) * ATT_D;
  float* lse_b = lse_out + ((long long)b * Hq + qh) * S;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int prow = 32 * w + (r & 3) + 8 * (r >> 2) + 4 * half;
    const int qrow = qbase + prow;
    float lv = l_lds[prow];
    float inv_l = (lv > 0.f) ? 1.f / lv : 0.f;
    unsigned short* orow = Ob + (long long)qrow * q_rowstride;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt)
      orow[nt * 32 + col] = f2bf(o_acc[nt][r] * inv_l);
    if (col == 0)
      lse_b[qrow] = (lv > 0.f) ? m_lds[prow] + __logf(lv) : -INFINITY;
  }
}

extern "C" __global__ void attn_fwd_swapped_kernel(
    const unsigned short*, const unsigned short*, const unsigned short*,
    unsigned short*, float*, int, int, int, int, float, int);

extern "C" void attn_fwd_v3_launch(const void* Q, const void* K,
                                   const void* V, void* O, float* lse, int B,
                                   int S, int Hq, int Hkv, float scale,
                                   bool causal, hipStream_t stream);

extern "C" void attn_fwd_launch(const void* Q, const void* K, const void* V,
                                void* O, float* lse, int B, int S, int Hq,
                                int Hkv, float scale, bool causal,
                                hipStream_t stream) {
  // NOTE: attn_fwd_kernel32 (32x32x16 MFMA, 128-row blocks) measures
  // SLOWER than this 16x16 kernel (94-97 vs 207 TF/s at the bench
  // shape) despite 2x the arithmetic intensity.  Measured eliminations
  // (round 1): removing the post-softmax barrier and single-writer LDS
  // m/l: null.  m/l in registers forces 1 wave/SIMD (VGPR cap): 63
  // TF/s.  So neither barriers nor softmax-state round-trips are the
  // bottleneck; the remaining suspects are the scalar V^T/P LDS
  // traffic and plain wave-level MFMA/VALU interleaving, which only
  // the full swapped-QK^T schedule (S^T = K Q^T, in-register column
  // softmax via permlane32_swap — semantics verified in
  // profiles/r01_hw_probe_semantics.txt) restructures away.  Dispatch
  // stays on the 16x16 kernel; SKY_ATTN_FWD_32=1 flips it for
  // experiments.
  // Swapped-operand kernel is the default (238 vs 207 TF/s measured at
  // the bench shape); SKY_ATTN_FWD_SWAPPED=0 falls back for A/B tests.
  // v3 (8-wave 32x32 deep-pipelined swapped schedule) is the round-2
  // default for the flagship shapes; SKY_ATTN_FWD_V3=0 for A/B.
  static const int use_v3 = [] {
    const char* e = getenv("SKY_ATTN_FWD_V3");
    return e ? atoi(e) : 1;
  }();
  if (use_v3 && S % 256 == 0) {
    attn_fwd_v3_launch(Q, K, V, O, lse, B, S, Hq, Hkv, scale, causal,
                       stream);
    return;
  }
  static const int use_swapped = [] {
    const char* e = getenv("SKY_ATTN_FWD_SWAPPED");
    return e ? atoi(e) : 1;
  }();
  if (use_swapped && S % BM == 0) {
    dim3 grid(S / BM, B * Hq);
    hipLaunchKernelGGL(attn_fwd_swapped_kernel, grid, dim3(256), 0,
                       stream, (const unsigned short*)Q,
                       (const unsigned short*)K, (const unsigned short*)V,
                       (unsigned short*)O, lse, B, S, Hq, Hkv, scale,
                       causal ? 1 : 0);
    return;
  }
  static const int use32 = [] {
    const char* e = getenv("SKY_ATTN_FWD_32");
    return e ? atoi(e) : 0;
  }();
  if (use32 && S % BM32 == 0) {
    dim3 grid(S / BM32, B * Hq);
    hipLaunchKernelGGL(attn_fwd_kernel32, grid, dim3(256), 0, stream,
                       (const unsigned short*)Q, (const unsigned short*)K,
                       (const unsigned short*)V, (unsigned short*)O, lse, B,
                       S, Hq, Hkv, scale, causal ? 1 : 0);
    return;
  }
  dim3 grid(S / BM, B * Hq);
  hipLaunchKernelGGL(attn_fwd_kernel, grid, dim3(256), 0, stream,
                     (const unsigned short*)Q, (const unsigned short*)K,
                     (const unsigned short*)V, (unsigned short*)O, lse, B, S,
                     Hq, Hkv, scale, causal ? 1 : 0);
}

// ---------------------------------------------------------------------------
// Swapped-operand forward: S^T = K Q^T, in-register column softmax,
// P^T redistributed to B-fragments with cross-lane shuffles, O^T
// accumulated and transposed on store.  Eliminates the P LDS roundtrip
// (32 swzP writes + 16 reads per wave-tile) and all softmax LDS state.
// Same MFMA count and the SAME k_lds/vt_lds reads as attn_fwd_kernel —
// only the operand roles change (K becomes A, Q becomes B), which is
// what makes the softmax land on lanes instead of register rows.
// permlane/tr probe groundwork: profiles/r01_hw_probe_semantics.txt.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256, 2) void attn_fwd_swapped_kernel(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, unsigned short* __restrict__ O,
    float* __restrict__ lse_out, int B, int S, int Hq, int Hkv, float scale,
    int causal) {
  __shared__ unsigned short k_lds[BN * ATT_D];    // [kv][d], swizzled, 16KB
  __shared__ unsigned short vt_lds[ATT_D * BN];   // [d][kv], swizzled, 16KB

  const int qt = gridDim.x - 1 - blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int qh = bh % Hq;
  const int kvh = qh / (Hq / Hkv);
  const int qbase = qt * BM;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;               // wave id: q rows [16w, 16w+16)
  const int lrow = lane & 15;           // q column owned by this lane
  const int lgrp = lane >> 4;

  const long long q_rowstride = (long long)Hq * ATT_D;
  const long long kv_rowstride = (long long)Hkv * ATT_D;
  const unsigned short* Qb = Q + ((long long)b * S * Hq + qh) * ATT_D;
  const unsigned short* Kb = K + ((long long)b * S * Hkv + kvh) * ATT_D;
  const unsigned short* Vb = V + ((long long)b * S * Hkv + kvh) * ATT_D;

  // Q as B-fragments: B[k=d][n=q]: lane holds Q[q=lrow][d=ks*32+lgrp*8+j]
  // (byte-identical to the A-fragment loads of the unswapped kernel).
  s16x8 q_b[4];
  {
    const int qrow = qbase + 16 * w + lrow;
    const unsigned short* src = Qb + (long long)qrow * q_rowstride;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks)
      q_b[ks] = *(const s16x8*)(src + ks * 32 + lgrp * 8);
  }

  // Per-lane softmax state for q column lrow (replicated across lgrp).
  float m_run = -INFINITY, l_run = 0.f;
  f32x4 o_t[8];  // O^T: C[row=d=ct*16+lgrp*4+r][col=q=lrow]
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) o_t[ct] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int n_kv_tiles = causal ? (qbase + BM + BN - 1) / BN : (S + BN - 1) / BN;

  s16x8 kpre[4], vpre[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int s_ch = (tid & 7) | ((i & 1) << 3);
    int s_row = (tid >> 3) | ((i >> 1) << 5);
    long long r = (long long)s_row * kv_rowstride + s_ch * 8;
    kpre[i] = *(const s16x8*)(Kb + r);
    vpre[i] = *(const s16x8*)(Vb + r);
  }

  const int my_q = qbase + 16 * w + lrow;
  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    __syncthreads();
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int s_ch = (tid & 7) | ((i & 1) << 3);
      int s_row = (tid >> 3) | ((i >> 1) << 5);
      *(s16x8*)((char*)k_lds + swz(s_row * 256 + s_ch * 16, s_row)) =
          kpre[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int d = s_ch * 8 + j;
        *(unsigned short*)((char*)vt_lds + swzT(d * 128 + s_row * 2, d)) =
            (unsigned short)vpre[i][j];
      }
    }
    __syncthreads();
    if (kt + 1 < n_kv_tiles) {
      const long long base = (long long)(kt + 1) * BN * kv_rowstride;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int s_ch = (tid & 7) | ((i & 1) << 3);
        int s_row = (tid >> 3) | ((i >> 1) << 5);
        long long r = base + (long long)s_row * kv_rowstride + s_ch * 8;
        kpre[i] = *(const s16x8*)(Kb + r);
        vpre[i] = *(const s16x8*)(Vb + r);
      }
    }
    const int kvbase = kt * BN;

    // ---- S^T = scale * K Q^T : 64 kv rows x 16 q cols for this wave.
    // ks outer / kv4 inner: 4 independent accumulator chains in flight
    // per MFMA latency window (kv4-outer serializes each 4-deep chain).
    f32x4 st_acc[4];
#pragma unroll
    for (int kv4 = 0; kv4 < 4; ++kv4)
      st_acc[kv4] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ks = 0; ks < 4; ++ks)
#pragma unroll
      for (int kv4 = 0; kv4 < 4; ++kv4) {
        int krow = kv4 * 16 + lrow;
        s16x8 a_k = *(const s16x8*)((char*)k_lds +
            swz(krow * 256 + (ks * 32 + lgrp * 8) * 2, krow));
        st_acc[kv4] = MFMA_BF16(as_bf16x8(a_k), as_bf16x8(q_b[ks]),
                                st_acc[kv4]);
      }

    // ---- mask + in-register column softmax (q = lrow on every lane).
    const bool need_mask =
        (causal && kvbase + BN - 1 > qbase + 16 * w) || (kvbase + BN > S);
    float mx = -INFINITY;
#pragma unroll
    for (int kv4 = 0; kv4 < 4; ++kv4)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float s = st_acc[kv4][r] * scale;
        if (need_mask) {
          int kv = kvbase + kv4 * 16 + lgrp * 4 + r;
          if ((causal && kv > my_q) || kv >= S) s = -INFINITY;
        }
        st_acc[kv4][r] = s;
        mx = fmaxf(mx, s);
      }
    mx = fmaxf(mx, __shfl_xor(mx, 16, 64));
    mx = fmaxf(mx, __shfl_xor(mx, 32, 64));
    float m_new = fmaxf(m_run, mx);
    float m_safe = (m_new == -INFINITY) ? 0.f : m_new;
    float corr = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_safe);
    m_run = m_new;
    float rs = 0.f;
#pragma unroll
    for (int kv4 = 0; kv4 < 4; ++kv4)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float e = (st_acc[kv4][r] == -INFINITY)
                      ? 0.f : __expf(st_acc[kv4][r] - m_safe);
        st_acc[kv4][r] = e;  // P^T stays in the S registers
        rs += e;
      }
    rs += __shfl_xor(rs, 16, 64);
    rs += __shfl_xor(rs, 32, 64);
    l_run = l_run * corr + rs;
#pragma unroll
    for (int ct = 0; ct < 8; ++ct)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_t[ct][r] *= corr;

    // ---- P^T -> B-fragments via cross-lane shuffles.
    // Dest (group g, elem j) needs P^T[kv = ks*32 + g*8 + j][q=lrow],
    // held by lane (2*(g&1) + (j>>2))*16 + lrow in sub-register
    // [kvsub = 2ks + (g>>1)][r = j&3].  kvsub depends on the dest group
    // (runtime), so shuffle both candidates (compile-time indices) and
    // select — 2 shfl + 1 cndmask per element.
    s16x8 p_b[2];
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int src = (2 * (lgrp & 1) + (j >> 2)) * 16 + lrow;
        float v0 = __shfl(st_acc[2 * ks][j & 3], src, 64);
        float v1 = __shfl(st_acc[2 * ks + 1][j & 3], src, 64);
        p_b[ks][j] = (short)f2bf_trunc((lgrp >> 1) ? v1 : v0);
      }
    }

    // ---- O^T += V^T P^T : A from vt_lds (same reads as unswapped).
#pragma unroll
    for (int ct = 0; ct < 8; ++ct)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        int vrow = ct * 16 + lrow;
        s16x8 a_vt = *(const s16x8*)((char*)vt_lds +
            swzT(vrow * 128 + (ks * 32 + lgrp * 8) * 2, vrow));
        o_t[ct] = MFMA_BF16(as_bf16x8(a_vt), as_bf16x8(p_b[ks]), o_t[ct]);
      }
  }

  // ---- epilogue: O[q][d] = O^T[d][q] / l (s16x4 stores, d block
  // ct*16 + lgrp*4), lse once per q column.
  unsigned short* Ob = O + ((long long)b * S * Hq + qh) * ATT_D;
  float* lse_b = lse_out + ((long long)b * Hq + qh) * S;
  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
  unsigned short* orow = Ob + (long long)my_q * q_rowstride;
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) {
    s16x4 ov;
#pragma unroll
    for (int r = 0; r < 4; ++r) ov[r] = (short)f2bf(o_t[ct][r] * inv_l);
    *(s16x4*)(orow + ct * 16 + lgrp * 4) = ov;
  }
  if (lgrp == 0)
    lse_b[my_q] = (l_run > 0.f) ? m_run + __logf(l_run) : -INFINITY;
}
