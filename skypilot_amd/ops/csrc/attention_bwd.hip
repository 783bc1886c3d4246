// Flash-attention backward (causal, GQA, D=128, bf16) for CDNA4/gfx950.
//
// Three kernels, no atomics (FA2-style split):
//   1. attn_bwd_pre:  Dvec = rowsum(dO * O)               [B,Hq,S] fp32
//   2. attn_bwd_dkv:  one block per kv tile; loops grouped q-heads and
//      q tiles >= diagonal; accumulates dK, dV in registers.
//   3. attn_bwd_dq:   one block per q tile; loops kv tiles <= diagonal.
// All MFMA B-fragments come from swizzled LDS tiles staged per iteration
// (normal + transposed copies where the contraction axis demands it).
//
// Math (P normalized via saved lse): P = exp(scale*QK^T - lse);
// dV = P^T dO; dP = dO V^T; dS = P*(dP - Dvec); dQ = scale*dS*K;
// dK = scale*dS^T*Q.
#include <cstdlib>
#include "common.h"

#define ATT_D 128
#define BM 64
#define BN 64

// ---------------------------------------------------------------------------
// Dvec preprocess: one wave per (b, qh, s) row.
// ---------------------------------------------------------------------------
extern "C" __global__ void attn_bwd_pre_kernel(
    const unsigned short* __restrict__ dO, const unsigned short* __restrict__ O,
    float* __restrict__ Dvec, long long rows, int Hq) {
  (void)Hq;
  const int lane = threadIdx.x & 63;
  const long long wave_id =
      (long long)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const long long stride = (long long)gridDim.x * (blockDim.x >> 6);
  for (long long row = wave_id; row < rows; row += stride) {
    // row = (b*S + s)*Hq + h ; layout [B,S,Hq,D] is contiguous in rows*D.
    const unsigned short* drow = dO + row * ATT_D;
    const unsigned short* orow = O + row * ATT_D;
    float acc = bf2f(drow[lane]) * bf2f(orow[lane]) +
                bf2f(drow[lane + 64]) * bf2f(orow[lane + 64]);
    acc = wave_reduce_sum(acc);
    // Dvec is stored [B,S,Hq] (same row order as the [B,S,Hq,D] tensors);
    // the dkv/dq kernels index it with stride Hq accordingly.
    if (lane == 0) Dvec[row] = acc;
  }
}

// ---------------------------------------------------------------------------
// dK/dV kernel. Grid: (S/BN, B*Hkv). LDS: Q,QT,dO,dOT (16KB each) + 8KB
// shared P/dS staging = 72KB -> 2 blocks/CU.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256, 2) void attn_bwd_dkv_kernel(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, const unsigned short* __restrict__ dO,
    const float* __restrict__ lse, const float* __restrict__ Dvec,
    unsigned short* __restrict__ dK, unsigned short* __restrict__ dV, int B,
    int S, int Hq, int Hkv, float scale, int causal) {
  __shared__ unsigned short q_lds[BM * ATT_D];
  __shared__ unsigned short qt_lds[ATT_D * BM];
  __shared__ unsigned short do_lds[BM * ATT_D];
  __shared__ unsigned short dot_lds[ATT_D * BM];
  __shared__ unsigned short p_lds[BN * BM];   // PT
  __shared__ unsigned short ds_lds[BN * BM];  // dST (separate so both
                                              // mfma passes share 1 barrier)

  const int kt = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hkv;
  const int kvh = bh % Hkv;
  const int group = Hq / Hkv;
  const int kvbase = kt * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;    // wave owns kv rows [16w, 16w+16)
  const int lrow = lane & 15;
  const int lgrp = lane >> 4;

  const long long q_rowstride = (long long)Hq * ATT_D;
  const long long kv_rowstride = (long long)Hkv * ATT_D;
  const unsigned short* Kb = K + ((long long)b * S * Hkv + kvh) * ATT_D;
  const unsigned short* Vb = V + ((long long)b * S * Hkv + kvh) * ATT_D;

  // K,V fragments for this wave's 16 kv rows (A-operand, fixed).
  s16x8 a_k[4], a_v[4];
  {
    const int kvrow = kvbase + 16 * w + lrow;
    const unsigned short* ksrc = Kb + (long long)kvrow * kv_rowstride;
    const unsigned short* vsrc = Vb + (long long)kvrow * kv_rowstride;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      a_k[ks] = *(const s16x8*)(ksrc + ks * 32 + lgrp * 8);
      a_v[ks] = *(const s16x8*)(vsrc + ks * 32 + lgrp * 8);
    }
  }

  f32x4 dv_acc[8], dk_acc[8];
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) {
    dv_acc[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
    dk_acc[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  for (int g = 0; g < group; ++g) {
    const int qh = kvh * group + g;
    const unsigned short* Qb = Q + ((long long)b * S * Hq + qh) * ATT_D;
    const unsigned short* dOb = dO + ((long long)b * S * Hq + qh) * ATT_D;
    const float* lse_b = lse + ((long long)b * Hq + qh) * S;
    // Dvec stored [B,S,Hq] (see pre kernel).
    const float* dvec_b = Dvec + (long long)b * S * Hq + qh;

    const int qt0 = causal ? kvbase / BM : 0;
    for (int qt = qt0; qt < S / BM; ++qt) {
      const int qbase = qt * BM;
      __syncthreads();
      // Stage Q, QT, dO, dOT (swizzled).
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        // 8-row x 8-chunk staging map: scatter writes hit all 32 banks
        // (see attn_fwd staging comment).
        int ch = (tid & 7) | ((i & 1) << 3);
        int row = (tid >> 3) | ((i >> 1) << 5);
        s16x8 qv = *(const s16x8*)(Qb + (long long)(qbase + row) * q_rowstride + ch * 8);
        *(s16x8*)((char*)q_lds + swz(row * 256 + ch * 16, row)) = qv;
        s16x8 dov = *(const s16x8*)(dOb + (long long)(qbase + row) * q_rowstride + ch * 8);
        *(s16x8*)((char*)do_lds + swz(row * 256 + ch * 16, row)) = dov;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int d = ch * 8 + j;
          *(unsigned short*)((char*)qt_lds + swzT(d * 128 + row * 2, d)) =
              (unsigned short)qv[j];
          *(unsigned short*)((char*)dot_lds + swzT(d * 128 + row * 2, d)) =
              (unsigned short)dov[j];
        }
      }
      __syncthreads();

      // ST = K Q^T (raw);  [kv 16][q 64] per wave.
      f32x4 st[4], dpt[4];
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        st[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
        dpt[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
          int qrow = ct * 16 + lrow;
          s16x8 bq = *(const s16x8*)((char*)q_lds +
                                     swz(qrow * 256 + (ks * 32 + lgrp * 8) * 2, qrow));
          st[ct] = MFMA_BF16(as_bf16x8(a_k[ks]), as_bf16x8(bq), st[ct]);
          s16x8 bdo = *(const s16x8*)((char*)do_lds +
                                      swz(qrow * 256 + (ks * 32 + lgrp * 8) * 2, qrow));
          dpt[ct] = MFMA_BF16(as_bf16x8(a_v[ks]), as_bf16x8(bdo), dpt[ct]);
        }
      }

      // PT = exp(scale*ST - lse[q]); dST = PT * (dPT - Dvec[q]).
      // Streamed straight to LDS (no pt/dst register arrays — keeping
      // them cost 32 VGPRs and dropped occupancy to 1 wave/SIMD).
      const int my_kvrow = kvbase + 16 * w + lgrp * 4;  // + r
      // No barrier needed before the P/dS writes: the prior iteration's
      // dV/dK reads of these buffers completed before this iteration's
      // top-of-loop barrier (two barriers ago).
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        int qcol = qbase + ct * 16 + lrow;
        float l = lse_b[qcol];
        float dv = dvec_b[(long long)qcol * Hq];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float sv = st[ct][r] * scale;
          float pv;
          if ((causal && qcol < my_kvrow + r) || l == -INFINITY)
            pv = 0.f;
          else
            pv = __expf(sv - l);
          int prow = 16 * w + lgrp * 4 + r;
          int pcol = ct * 16 + lrow;
          int off = swzP(prow * 128 + pcol * 2, prow);
          *(unsigned short*)((char*)p_lds + off) = f2bf_trunc(pv);
          *(unsigned short*)((char*)ds_lds + off) =
              f2bf(pv * (dpt[ct][r] - dv));
        }
      }
      __syncthreads();

      // dV += PT * dO (via dOT);  dK += dST * Q (via QT).
#pragma unroll
      for (int ct = 0; ct < 8; ++ct)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          int prow = 16 * w + lrow;
          int a_off = swzP(prow * 128 + (ks * 32 + lgrp * 8) * 2, prow);
          int brow = ct * 16 + lrow;
          int b_off = swzT(brow * 128 + (ks * 32 + lgrp * 8) * 2, brow);
          s16x8 pfrag = *(const s16x8*)((char*)p_lds + a_off);
          s16x8 dofrag = *(const s16x8*)((char*)dot_lds + b_off);
          dv_acc[ct] = MFMA_BF16(as_bf16x8(pfrag), as_bf16x8(dofrag),
                                 dv_acc[ct]);
          s16x8 dsfrag = *(const s16x8*)((char*)ds_lds + a_off);
          s16x8 qfrag = *(const s16x8*)((char*)qt_lds + b_off);
          dk_acc[ct] = MFMA_BF16(as_bf16x8(dsfrag), as_bf16x8(qfrag),
                                 dk_acc[ct]);
        }
    }
  }

  // Write dK, dV (bf16), applying scale to dK.
  unsigned short* dKb = dK + ((long long)b * S * Hkv + kvh) * ATT_D;
  unsigned short* dVb = dV + ((long long)b * S * Hkv + kvh) * ATT_D;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kvrow = kvbase + 16 * w + lgrp * 4 + r;
    unsigned short* krow = dKb + (long long)kvrow * kv_rowstride;
    unsigned short* vrow = dVb + (long long)kvrow * kv_rowstride;
#pragma unroll
    for (int ct = 0; ct < 8; ++ct) {
      krow[ct * 16 + lrow] = f2bf(dk_acc[ct][r] * scale);
      vrow[ct * 16 + lrow] = f2bf(dv_acc[ct][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// dQ kernel. Grid: (S/BM, B*Hq). LDS: K + KT + V + dS = 56KB.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256, 1) void attn_bwd_dq_kernel(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, const unsigned short* __restrict__ dO,
    const float* __restrict__ lse, const float* __restrict__ Dvec,
    unsigned short* __restrict__ dQ, int B, int S, int Hq, int Hkv,
    float scale, int causal) {
  __shared__ unsigned short k_lds[BN * ATT_D];
  __shared__ unsigned short kt_lds[ATT_D * BN];
  __shared__ unsigned short v_lds[BN * ATT_D];
  __shared__ unsigned short ds_lds[BM * BN];

  // Heavy blocks first (see attn_fwd).
  const int qt = gridDim.x - 1 - blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int qh = bh % Hq;
  const int kvh = qh / (Hq / Hkv);
  const int qbase = qt * BM;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int lrow = lane & 15;
  const int lgrp = lane >> 4;

  const long long q_rowstride = (long long)Hq * ATT_D;
  const long long kv_rowstride = (long long)Hkv * ATT_D;
  const unsigned short* Qb = Q + ((long long)b * S * Hq + qh) * ATT_D;
  const unsigned short* Kb = K + ((long long)b * S * Hkv + kvh) * ATT_D;
  const unsigned short* Vb = V + ((long long)b * S * Hkv + kvh) * ATT_D;
  const unsigned short* dOb = dO + ((long long)b * S * Hq + qh) * ATT_D;
  const float* lse_b = lse + ((long long)b * Hq + qh) * S;
  const float* dvec_b = Dvec + (long long)b * S * Hq + qh;

  // Fixed A fragments: Q rows and dO rows for this wave.
  s16x8 a_q[4], a_do[4];
  float my_lse[4], my_dvec[4];
  {
    const int qrow = qbase + 16 * w + lrow;
    const unsigned short* qsrc = Qb + (long long)qrow * q_rowstride;
    const unsigned short* dsrc = dOb + (long long)qrow * q_rowstride;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      a_q[ks] = *(const s16x8*)(qsrc + ks * 32 + lgrp * 8);
      a_do[ks] = *(const s16x8*)(dsrc + ks * 32 + lgrp * 8);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int rr = qbase + 16 * w + lgrp * 4 + r;
      my_lse[r] = lse_b[rr];
      my_dvec[r] = dvec_b[(long long)rr * Hq];
    }
  }

  f32x4 dq_acc[8];
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) dq_acc[ct] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int n_kv_tiles = causal ? (qbase + BM + BN - 1) / BN : S / BN;
  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int kvbase = kt * BN;
    __syncthreads();
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int ch = (tid & 7) | ((i & 1) << 3);
      int row = (tid >> 3) | ((i >> 1) << 5);
      s16x8 kv8 = *(const s16x8*)(Kb + (long long)(kvbase + row) * kv_rowstride + ch * 8);
      *(s16x8*)((char*)k_lds + swz(row * 256 + ch * 16, row)) = kv8;
      s16x8 vv8 = *(const s16x8*)(Vb + (long long)(kvbase + row) * kv_rowstride + ch * 8);
      *(s16x8*)((char*)v_lds + swz(row * 256 + ch * 16, row)) = vv8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int d = ch * 8 + j;
        *(unsigned short*)((char*)kt_lds + swzT(d * 128 + row * 2, d)) =
            (unsigned short)kv8[j];
      }
    }
    __syncthreads();

    // S = Q K^T (raw), dP = dO V^T; both [q 16][kv 64] per wave.
    f32x4 s_acc[4], dp[4];
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
      s_acc[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
      dp[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        int krow = ct * 16 + lrow;
        s16x8 bk = *(const s16x8*)((char*)k_lds +
                                   swz(krow * 256 + (ks * 32 + lgrp * 8) * 2, krow));
        s_acc[ct] = MFMA_BF16(as_bf16x8(a_q[ks]), as_bf16x8(bk), s_acc[ct]);
        s16x8 bv = *(const s16x8*)((char*)v_lds +
                                   swz(krow * 256 + (ks * 32 + lgrp * 8) * 2, krow));
        dp[ct] = MFMA_BF16(as_bf16x8(a_do[ks]), as_bf16x8(bv), dp[ct]);
      }
    }

    // dS = P * (dP - Dvec);  P = exp(scale*S - lse).
    // (No barrier: prior-iteration dQ reads of ds_lds finished before the
    // top-of-loop barrier.)
    const int my_qrow = qbase + 16 * w + lgrp * 4;
#pragma unroll
    for (int ct = 0; ct < 4; ++ct)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int col = kvbase + ct * 16 + lrow;
        float pv;
        if ((causal && col > my_qrow + r) || my_lse[r] == -INFINITY)
          pv = 0.f;
        else
          pv = __expf(s_acc[ct][r] * scale - my_lse[r]);
        float ds = pv * (dp[ct][r] - my_dvec[r]);
        int prow = 16 * w + lgrp * 4 + r;
        int pcol = ct * 16 + lrow;
        *(unsigned short*)((char*)ds_lds + swzP(prow * 128 + pcol * 2, prow)) =
            f2bf(ds);  // dS can be negative/large: keep rounded convert
      }
    __syncthreads();

    // dQ += dS K (via KT).
#pragma unroll
    for (int ct = 0; ct < 8; ++ct)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        int prow = 16 * w + lrow;
        s16x8 afrag = *(const s16x8*)((char*)ds_lds +
                                      swzP(prow * 128 + (ks * 32 + lgrp * 8) * 2, prow));
        int krow = ct * 16 + lrow;
        s16x8 bfrag = *(const s16x8*)((char*)kt_lds +
                                      swzT(krow * 128 + (ks * 32 + lgrp * 8) * 2, krow));
        dq_acc[ct] = MFMA_BF16(as_bf16x8(afrag), as_bf16x8(bfrag), dq_acc[ct]);
      }
  }

  unsigned short* dQb = dQ + ((long long)b * S * Hq + qh) * ATT_D;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = qbase + 16 * w + lgrp * 4 + r;
    unsigned short* orow = dQb + (long long)qrow * q_rowstride;
#pragma unroll
    for (int ct = 0; ct < 8; ++ct)
      orow[ct * 16 + lrow] = f2bf(dq_acc[ct][r] * scale);
  }
}


// ---------------------------------------------------------------------------
// 32x32x16 dK/dV kernel: 512 threads (8 waves), role-split per q-tile —
// waves 0-3 compute S^T tiles (+exp -> p_lds), waves 4-7 compute dP^T
// tiles (+dS -> ds_lds), then all 8 accumulate their own (kv-half,
// d-quarter) dV and dK 32x32 output tiles.  Same FLOPs as the 16x16
// kernel at half the instruction count and half the per-FLOP barrier
// cost; no online softmax (lse precomputed) so the 32x32 C-layout costs
// nothing here.  Layouts verified by test_mfma32_layout_probe.
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(16))) float f32x16;
#define MFMA32_BF16(a, b, c) \
  __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0)

extern "C" __global__ __launch_bounds__(512, 2) void attn_bwd_dkv_kernel32(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, const unsigned short* __restrict__ dO,
    const float* __restrict__ lse, const float* __restrict__ Dvec,
    unsigned short* __restrict__ dK, unsigned short* __restrict__ dV, int B,
    int S, int Hq, int Hkv, float scale, int causal) {
  __shared__ unsigned short q_lds[BM * ATT_D];
  __shared__ unsigned short qt_lds[ATT_D * BM];
  __shared__ unsigned short do_lds[BM * ATT_D];
  __shared__ unsigned short dot_lds[ATT_D * BM];
  __shared__ unsigned short p_lds[BN * BM];
  __shared__ unsigned short ds_lds[BN * BM];

  const int kt = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hkv;
  const int kvh_head = bh % Hkv;
  const int group = Hq / Hkv;
  const int kvbase = kt * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;                 // 8 waves
  const int col = lane & 31;
  const int half = lane >> 5;

  const long long q_rowstride = (long long)Hq * ATT_D;
  const long long kv_rowstride = (long long)Hkv * ATT_D;
  const unsigned short* Kb = K + ((long long)b * S * Hkv + kvh_head) * ATT_D;
  const unsigned short* Vb = V + ((long long)b * S * Hkv + kvh_head) * ATT_D;

  // Role assignment.
  const bool is_st = (w < 4);             // S^T waves vs dP^T waves
  const int t_kvh = is_st ? (w >> 1) : ((w - 4) >> 1);  // kv half (ST/dPT)
  const int t_qh = is_st ? (w & 1) : ((w - 4) & 1);     // q half (ST/dPT)
  const int o_kvh = w >> 2;               // output tile: kv half
  const int o_dq = w & 3;                 // output tile: d quarter

  // A-operand fragments held in registers: K rows (ST waves) or V rows
  // (dPT waves) for the wave's ST/dPT tile — fixed for the whole block.
  s16x8 a_kv[8];
  {
    const unsigned short* src =
        (is_st ? Kb : Vb) +
        (long long)(kvbase + t_kvh * 32 + col) * kv_rowstride;
#pragma unroll
    for (int ks = 0; ks < 8; ++ks)
      a_kv[ks] = *(const s16x8*)(src + ks * 16 + half * 8);
  }

  f32x16 dv_acc, dk_acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    dv_acc[r] = 0.f;
    dk_acc[r] = 0.f;
  }

  for (int g = 0; g < group; ++g) {
    const int qh_head = kvh_head * group + g;
    const unsigned short* Qb = Q + ((long long)b * S * Hq + qh_head) * ATT_D;
    const unsigned short* dOb =
        dO + ((long long)b * S * Hq + qh_head) * ATT_D;
    const float* lse_b = lse + ((long long)b * Hq + qh_head) * S;
    const float* dvec_b = Dvec + (long long)b * S * Hq + qh_head;

    const int qt0 = causal ? kvbase / BM : 0;
    for (int qt = qt0; qt < S / BM; ++qt) {
      const int qbase = qt * BM;
      __syncthreads();
      // Stage Q/QT/dO/dOT — 512 threads, 2 chunks each (same map as the
      // 16x16 kernel: ch=(t&7)|((i&1)<<3), row=(t>>3)|((i>>1)<<5) with
      // t = tid%256 and the high thread bit folded into i).
#pragma unroll
      for (int i2 = 0; i2 < 2; ++i2) {
        int i = i2 * 2 + (tid >> 8);      // 0..3
        int t = tid & 255;
        int ch = (t & 7) | ((i & 1) << 3);
        int row = (t >> 3) | ((i >> 1) << 5);
        s16x8 qv = *(const s16x8*)(Qb + (long long)(qbase + row) *
                                       q_rowstride + ch * 8);
        *(s16x8*)((char*)q_lds + swz(row * 256 + ch * 16, row)) = qv;
        s16x8 dov = *(const s16x8*)(dOb + (long long)(qbase + row) *
                                        q_rowstride + ch * 8);
        *(s16x8*)((char*)do_lds + swz(row * 256 + ch * 16, row)) = dov;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int d = ch * 8 + j;
          *(unsigned short*)((char*)qt_lds + swzT(d * 128 + row * 2, d)) =
              (unsigned short)qv[j];
          *(unsigned short*)((char*)dot_lds + swzT(d * 128 + row * 2, d)) =
              (unsigned short)dov[j];
        }
      }
      __syncthreads();

      // ST waves: st = K Q^T tile [kv 32][q 32]; dPT waves: dpt = V dO^T.
      const unsigned short* b_src = is_st ? q_lds : do_lds;
      f32x16 acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = 0.f;
#pragma unroll
      for (int ks = 0; ks < 8; ++ks) {
        int qrow = t_qh * 32 + col;
        s16x8 bfrag = *(const s16x8*)(
            (char*)b_src + swz(qrow * 256 + (ks * 16 + half * 8) * 2, qrow));
        acc = MFMA32_BF16(as_bf16x8(a_kv[ks]), as_bf16x8(bfrag), acc);
      }

      if (is_st) {
        // PT = exp(scale*ST - lse[q]) with causal mask -> p_lds.
        const int qcol = qbase + t_qh * 32 + col;
        const float l = lse_b[qcol];
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kvrow = t_kvh * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          float pv;
          if ((causal && qcol < kvbase + kvrow) || l == -INFINITY)
            pv = 0.f;
          else
            pv = __expf(acc[r] * scale - l);
          int prow = kvrow;
          int pcol = t_qh * 32 + col;
          *(unsigned short*)((char*)p_lds +
                             swzP(prow * 128 + pcol * 2, prow)) =
              f2bf_trunc(pv);
        }
      }
      __syncthreads();

      if (!is_st) {
        // dST = PT * (dPT - Dvec[q]) -> ds_lds (reads PT written above).
        const int qcol = qbase + t_qh * 32 + col;
        const float dvq = dvec_b[(long long)qcol * Hq];
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int kvrow = t_kvh * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          int off = swzP(kvrow * 128 + (t_qh * 32 + col) * 2, kvrow);
          float pv = bf2f(*(const unsigned short*)((char*)p_lds + off));
          *(unsigned short*)((char*)ds_lds + off) =
              f2bf(pv * (acc[r] - dvq));
        }
      }
      // dV += PT x dO (A from p_lds, B from dot_lds) — PT is ready for
      // all waves after the barrier above.
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        int arow = o_kvh * 32 + col;
        s16x8 afrag = *(const s16x8*)(
            (char*)p_lds +
            swzP(arow * 128 + (ks * 16 + half * 8) * 2, arow));
        int brow = o_dq * 32 + col;
        s16x8 bfrag = *(const s16x8*)(
            (char*)dot_lds +
            swzT(brow * 128 + (ks * 16 + half * 8) * 2, brow));
        dv_acc = MFMA32_BF16(as_bf16x8(afrag), as_bf16x8(bfrag), dv_acc);
      }
      __syncthreads();
      // dK += dST x Q (A from ds_lds, B from qt_lds).
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        int arow = o_kvh * 32 + col;
        s16x8 afrag = *(const s16x8*)(
            (char*)ds_lds +
            swzP(arow * 128 + (ks * 16 + half * 8) * 2, arow));
        int brow = o_dq * 32 + col;
        s16x8 bfrag = *(const s16x8*)(
            (char*)qt_lds +
            swzT(brow * 128 + (ks * 16 + half * 8) * 2, brow));
        dk_acc = MFMA32_BF16(as_bf16x8(afrag), as_bf16x8(bfrag), dk_acc);
      }
    }
  }

  // Write this wave's dK/dV 32x32 tiles (bf16; scale folded into dK).
  unsigned short* dKb = dK + ((long long)b * S * Hkv + kvh_head) * ATT_D;
  unsigned short* dVb = dV + ((long long)b * S * Hkv + kvh_head) * ATT_D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kvrow =
        kvbase + o_kvh * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
    const int d = o_dq * 32 + col;
    dKb[(long long)kvrow * kv_rowstride + d] = f2bf(dk_acc[r] * scale);
    dVb[(long long)kvrow * kv_rowstride + d] = f2bf(dv_acc[r]);
  }
}

extern "C" __global__ void attn_bwd_dkv_tr_kernel(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const unsigned short*, const float*, const float*, unsigned short*,
    unsigned short*, int, int, int, int, float, int);

extern "C" void attn_bwd_v3_launch(const void* Q, const void* K,
                                   const void* V, const void* dO,
                                   const float* lse, const float* Dvec,
                                   void* dQ, void* dK, void* dV, int B,
                                   int S, int Hq, int Hkv, float scale,
                                   bool causal, hipStream_t stream);

extern "C" __global__ void attn_bwd_dkv_swapped_kernel(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const unsigned short*, const float*, const float*, unsigned short*,
    unsigned short*, int, int, int, int, float, int);

extern "C" __global__ void attn_bwd_dq_swapped_kernel(
    const unsigned short*, const unsigned short*, const unsigned short*,
    const unsigned short*, const float*, const float*, unsigned short*,
    int, int, int, int, float, int);

extern "C" void attn_bwd_launch(const void* Q, const void* K, const void* V,
                                const void* O, const void* dO,
                                const float* lse, float* Dvec, void* dQ,
                                void* dK, void* dV, int B, int S, int Hq,
                                int Hkv, float scale, bool causal,
                                hipStream_t stream) {
  long long rows = (long long)B * S * Hq;
  hipLaunchKernelGGL(attn_bwd_pre_kernel, dim3(membound_grid(rows, 4)),
                     dim3(256), 0, stream, (const unsigned short*)dO,
                     (const unsigned short*)O, Dvec, rows, Hq);
  // v3 (32x32 deep-pipelined dkv/dq) is the round-2 default for the
  // flagship shapes; SKY_ATTN_BWD_V3=0 for A/B.
  static const int use_bwd_v3 = [] {
    const char* e = getenv("SKY_ATTN_BWD_V3");
    return e ? atoi(e) : 1;
  }();
  if (use_bwd_v3 && S % 128 == 0) {
    attn_bwd_v3_launch(Q, K, V, dO, lse, Dvec, dQ, dK, dV, B, S, Hq, Hkv,
                       scale, causal, stream);
    return;
  }
  // NOTE: attn_bwd_dkv_kernel32 (8-wave 32x32, half the instructions)
  // measured 148 vs 163 TF/s — its 512-thread block leaves 1 block/CU,
  // losing the cross-block barrier/compute overlap that two independent
  // 256-thread blocks provide.  Kept for the round-2 deep-pipeline
  // rewrite; dispatch stays on the 16x16 kernel.
  static const int dkv_swapped = [] {
    const char* e = getenv("SKY_ATTN_DKV_SWAPPED");
    return e ? atoi(e) : 0;
  }();
  static const int dkv_tr = [] {
    const char* e = getenv("SKY_ATTN_DKV_TR");
    return e ? atoi(e) : 0;
  }();
  if (dkv_tr)
    hipLaunchKernelGGL(attn_bwd_dkv_tr_kernel, dim3(S / BN, B * Hkv),
                       dim3(256), 0, stream, (const unsigned short*)Q,
                       (const unsigned short*)K, (const unsigned short*)V,
                       (const unsigned short*)dO, lse, Dvec,
                       (unsigned short*)dK, (unsigned short*)dV, B, S, Hq,
                       Hkv, scale, causal ? 1 : 0);
  else if (dkv_swapped)
    hipLaunchKernelGGL(attn_bwd_dkv_swapped_kernel, dim3(S / BN, B * Hkv),
                       dim3(256), 0, stream, (const unsigned short*)Q,
                       (const unsigned short*)K, (const unsigned short*)V,
                       (const unsigned short*)dO, lse, Dvec,
                       (unsigned short*)dK, (unsigned short*)dV, B, S, Hq,
                       Hkv, scale, causal ? 1 : 0);
  else
    hipLaunchKernelGGL(attn_bwd_dkv_kernel, dim3(S / BN, B * Hkv),
                       dim3(256), 0, stream, (const unsigned short*)Q,
                       (const unsigned short*)K, (const unsigned short*)V,
                       (const unsigned short*)dO, lse, Dvec,
                       (unsigned short*)dK, (unsigned short*)dV, B, S, Hq,
                       Hkv, scale, causal ? 1 : 0);
  // Swapped dq is the default (bwd 163 -> 169 TF/s measured);
  // SKY_ATTN_DQ_SWAPPED=0 falls back for A/B tests.
  static const int dq_swapped = [] {
    const char* e = getenv("SKY_ATTN_DQ_SWAPPED");
    return e ? atoi(e) : 1;
  }();
  if (dq_swapped)
    hipLaunchKernelGGL(attn_bwd_dq_swapped_kernel, dim3(S / BM, B * Hq),
                       dim3(256), 0, stream, (const unsigned short*)Q,
                       (const unsigned short*)K, (const unsigned short*)V,
                       (const unsigned short*)dO, lse, Dvec,
                       (unsigned short*)dQ, B, S, Hq, Hkv, scale,
                       causal ? 1 : 0);
  else
    hipLaunchKernelGGL(attn_bwd_dq_kernel, dim3(S / BM, B * Hq), dim3(256),
                       0, stream, (const unsigned short*)Q,
                       (const unsigned short*)K, (const unsigned short*)V,
                       (const unsigned short*)dO, lse, Dvec,
                       (unsigned short*)dQ, B, S, Hq, Hkv, scale,
                       causal ? 1 : 0);
}

// ---------------------------------------------------------------------------
// Swapped-operand dQ kernel (same trick as attn_fwd_swapped_kernel):
// S^T = K Q^T and dP^T = V dO^T put q on lanes, so P^T/dS^T stay in
// registers (lse/Dvec become per-lane scalars) and dQ^T = K^T dS^T
// consumes dS^T as shuffled B-fragments — the ds_lds roundtrip and one
// of the two barriers per tile disappear.
// ---------------------------------------------------------------------------
// (launch_bounds(256,3) forces 168 VGPRs + 100 B/lane scratch for 3
// waves/SIMD: measured 158 vs 169 TF/s — spills cost more than the
// extra wave hides.)
extern "C" __global__ __launch_bounds__(256, 1) void attn_bwd_dq_swapped_kernel(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, const unsigned short* __restrict__ dO,
    const float* __restrict__ lse, const float* __restrict__ Dvec,
    unsigned short* __restrict__ dQ, int B, int S, int Hq, int Hkv,
    float scale, int causal) {
  __shared__ unsigned short k_lds[BN * ATT_D];
  __shared__ unsigned short kt_lds[ATT_D * BN];
  __shared__ unsigned short v_lds[BN * ATT_D];

  const int qt = gridDim.x - 1 - blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int qh = bh % Hq;
  const int kvh = qh / (Hq / Hkv);
  const int qbase = qt * BM;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int lrow = lane & 15;
  const int lgrp = lane >> 4;

  const long long q_rowstride = (long long)Hq * ATT_D;
  const long long kv_rowstride = (long long)Hkv * ATT_D;
  const unsigned short* Qb = Q + ((long long)b * S * Hq + qh) * ATT_D;
  const unsigned short* Kb = K + ((long long)b * S * Hkv + kvh) * ATT_D;
  const unsigned short* Vb = V + ((long long)b * S * Hkv + kvh) * ATT_D;
  const unsigned short* dOb = dO + ((long long)b * S * Hq + qh) * ATT_D;
  const float* lse_b = lse + ((long long)b * Hq + qh) * S;
  const float* dvec_b = Dvec + (long long)b * S * Hq + qh;

  // Q / dO as B-fragments (byte-identical loads to the A-fragments of
  // the unswapped kernel); lse/D collapse to per-lane scalars.
  s16x8 q_b[4], do_b[4];
  const int my_q = qbase + 16 * w + lrow;
  {
    const unsigned short* qsrc = Qb + (long long)my_q * q_rowstride;
    const unsigned short* dsrc = dOb + (long long)my_q * q_rowstride;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      q_b[ks] = *(const s16x8*)(qsrc + ks * 32 + lgrp * 8);
      do_b[ks] = *(const s16x8*)(dsrc + ks * 32 + lgrp * 8);
    }
  }
  const float my_lse = lse_b[my_q];
  const float my_dvec = dvec_b[(long long)my_q * Hq];

  f32x4 dq_t[8];  // dQ^T: C[row=d=ct*16+lgrp*4+r][col=q=lrow]
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) dq_t[ct] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int n_kv_tiles = causal ? (qbase + BM + BN - 1) / BN : S / BN;
  for (int kt = 0; kt < n_kv_tiles; ++kt) {
    const int kvbase = kt * BN;
    __syncthreads();
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int ch = (tid & 7) | ((i & 1) << 3);
      int row = (tid >> 3) | ((i >> 1) << 5);
      s16x8 kv8 = *(const s16x8*)(Kb + (long long)(kvbase + row) * kv_rowstride + ch * 8);
      *(s16x8*)((char*)k_lds + swz(row * 256 + ch * 16, row)) = kv8;
      s16x8 vv8 = *(const s16x8*)(Vb + (long long)(kvbase + row) * kv_rowstride + ch * 8);
      *(s16x8*)((char*)v_lds + swz(row * 256 + ch * 16, row)) = vv8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int d = ch * 8 + j;
        *(unsigned short*)((char*)kt_lds + swzT(d * 128 + row * 2, d)) =
            (unsigned short)kv8[j];
      }
    }
    __syncthreads();

    // S^T = K Q^T, dP^T = V dO^T (interleaved independent chains).
    f32x4 st[4], dpt[4];
#pragma unroll
    for (int kv4 = 0; kv4 < 4; ++kv4) {
      st[kv4] = f32x4{0.f, 0.f, 0.f, 0.f};
      dpt[kv4] = f32x4{0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int ks = 0; ks < 4; ++ks)
#pragma unroll
      for (int kv4 = 0; kv4 < 4; ++kv4) {
        int krow = kv4 * 16 + lrow;
        s16x8 a_k = *(const s16x8*)((char*)k_lds +
            swz(krow * 256 + (ks * 32 + lgrp * 8) * 2, krow));
        st[kv4] = MFMA_BF16(as_bf16x8(a_k), as_bf16x8(q_b[ks]), st[kv4]);
        s16x8 a_v = *(const s16x8*)((char*)v_lds +
            swz(krow * 256 + (ks * 32 + lgrp * 8) * 2, krow));
        dpt[kv4] = MFMA_BF16(as_bf16x8(a_v), as_bf16x8(do_b[ks]),
                             dpt[kv4]);
      }

    // dS^T = P^T * (dP^T - D);  P^T = exp(scale*S^T - lse).
#pragma unroll
    for (int kv4 = 0; kv4 < 4; ++kv4)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int kv = kvbase + kv4 * 16 + lgrp * 4 + r;
        float pv;
        if ((causal && kv > my_q) || my_lse == -INFINITY)
          pv = 0.f;
        else
          pv = __expf(st[kv4][r] * scale - my_lse);
        st[kv4][r] = pv * (dpt[kv4][r] - my_dvec);
      }

    // dS^T -> B-fragments (same shuffle recipe as attn_fwd_swapped).
    s16x8 ds_b[2];
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int src = (2 * (lgrp & 1) + (j >> 2)) * 16 + lrow;
        float v0 = __shfl(st[2 * ks][j & 3], src, 64);
        float v1 = __shfl(st[2 * ks + 1][j & 3], src, 64);
        ds_b[ks][j] = (short)f2bf((lgrp >> 1) ? v1 : v0);
      }

    // dQ^T += K^T dS^T : A from kt_lds (same reads as unswapped).
#pragma unroll
    for (int ct = 0; ct < 8; ++ct)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        int krow = ct * 16 + lrow;
        s16x8 a_kt = *(const s16x8*)((char*)kt_lds +
            swzT(krow * 128 + (ks * 32 + lgrp * 8) * 2, krow));
        dq_t[ct] = MFMA_BF16(as_bf16x8(a_kt), as_bf16x8(ds_b[ks]),
                             dq_t[ct]);
      }
  }

  unsigned short* dQb = dQ + ((long long)b * S * Hq + qh) * ATT_D;
  unsigned short* orow = dQb + (long long)my_q * q_rowstride;
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) {
    s16x4 ov;
#pragma unroll
    for (int r = 0; r < 4; ++r)
      ov[r] = (short)f2bf(dq_t[ct][r] * scale);
    *(s16x4*)(orow + ct * 16 + lgrp * 4) = ov;
  }
}

// ---------------------------------------------------------------------------
// Swapped-operand dK/dV kernel: S = Q K^T / dP = dO V^T keep q in
// register rows and kv on lanes, so P and dS stay in registers and feed
// dV^T = dO^T P and dK^T = Q^T dS as shuffled B-fragments — the
// p_lds/ds_lds roundtrips and one barrier per q-tile disappear (LDS
// 80 -> 64 KB).  Q^T/dO^T staging is unchanged.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256, 2) void attn_bwd_dkv_swapped_kernel(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, const unsigned short* __restrict__ dO,
    const float* __restrict__ lse, const float* __restrict__ Dvec,
    unsigned short* __restrict__ dK, unsigned short* __restrict__ dV, int B,
    int S, int Hq, int Hkv, float scale, int causal) {
  __shared__ unsigned short q_lds[BM * ATT_D];
  __shared__ unsigned short qt_lds[ATT_D * BM];
  __shared__ unsigned short do_lds[BM * ATT_D];
  __shared__ unsigned short dot_lds[ATT_D * BM];

  const int kt = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hkv;
  const int kvh = bh % Hkv;
  const int group = Hq / Hkv;
  const int kvbase = kt * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;    // wave owns kv rows [16w, 16w+16)
  const int lrow = lane & 15;
  const int lgrp = lane >> 4;

  const long long q_rowstride = (long long)Hq * ATT_D;
  const long long kv_rowstride = (long long)Hkv * ATT_D;
  const unsigned short* Kb = K + ((long long)b * S * Hkv + kvh) * ATT_D;
  const unsigned short* Vb = V + ((long long)b * S * Hkv + kvh) * ATT_D;

  // K,V as B-fragments for this wave's 16 kv columns (byte-identical
  // loads to the unswapped kernel's A-fragments).
  s16x8 k_b[4], v_b[4];
  const int my_kv = kvbase + 16 * w + lrow;
  {
    const unsigned short* ksrc = Kb + (long long)my_kv * kv_rowstride;
    const unsigned short* vsrc = Vb + (long long)my_kv * kv_rowstride;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      k_b[ks] = *(const s16x8*)(ksrc + ks * 32 + lgrp * 8);
      v_b[ks] = *(const s16x8*)(vsrc + ks * 32 + lgrp * 8);
    }
  }

  f32x4 dv_t[8], dk_t[8];  // C[row=d=ct*16+lgrp*4+r][col=kv=lrow]
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) {
    dv_t[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
    dk_t[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  for (int g = 0; g < group; ++g) {
    const int qh = kvh * group + g;
    const unsigned short* Qb = Q + ((long long)b * S * Hq + qh) * ATT_D;
    const unsigned short* dOb = dO + ((long long)b * S * Hq + qh) * ATT_D;
    const float* lse_b = lse + ((long long)b * Hq + qh) * S;
    const float* dvec_b = Dvec + (long long)b * S * Hq + qh;

    const int qt0 = causal ? kvbase / BM : 0;
    for (int qt = qt0; qt < S / BM; ++qt) {
      const int qbase = qt * BM;
      __syncthreads();
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int ch = (tid & 7) | ((i & 1) << 3);
        int row = (tid >> 3) | ((i >> 1) << 5);
        s16x8 qv = *(const s16x8*)(Qb + (long long)(qbase + row) * q_rowstride + ch * 8);
        *(s16x8*)((char*)q_lds + swz(row * 256 + ch * 16, row)) = qv;
        s16x8 dov = *(const s16x8*)(dOb + (long long)(qbase + row) * q_rowstride + ch * 8);
        *(s16x8*)((char*)do_lds + swz(row * 256 + ch * 16, row)) = dov;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int d = ch * 8 + j;
          *(unsigned short*)((char*)qt_lds + swzT(d * 128 + row * 2, d)) =
              (unsigned short)qv[j];
          *(unsigned short*)((char*)dot_lds + swzT(d * 128 + row * 2, d)) =
              (unsigned short)dov[j];
        }
      }
      __syncthreads();

      // S = Q K^T, dP = dO V^T : [q 64 rows][kv 16 lanes] per wave.
      f32x4 sp[4], dp[4];
#pragma unroll
      for (int qsub = 0; qsub < 4; ++qsub) {
        sp[qsub] = f32x4{0.f, 0.f, 0.f, 0.f};
        dp[qsub] = f32x4{0.f, 0.f, 0.f, 0.f};
      }
#pragma unroll
      for (int ks = 0; ks < 4; ++ks)
#pragma unroll
        for (int qsub = 0; qsub < 4; ++qsub) {
          int qrow = qsub * 16 + lrow;
          s16x8 a_q = *(const s16x8*)((char*)q_lds +
              swz(qrow * 256 + (ks * 32 + lgrp * 8) * 2, qrow));
          sp[qsub] = MFMA_BF16(as_bf16x8(a_q), as_bf16x8(k_b[ks]),
                               sp[qsub]);
          s16x8 a_do = *(const s16x8*)((char*)do_lds +
              swz(qrow * 256 + (ks * 32 + lgrp * 8) * 2, qrow));
          dp[qsub] = MFMA_BF16(as_bf16x8(a_do), as_bf16x8(v_b[ks]),
                               dp[qsub]);
        }

      // P = exp(scale*S - lse[q]); dS = P * (dP - D[q]); in registers.
      // lse/D loaded once per lane column (4 each) and redistributed
      // to register rows by shuffle — 16 scalar loads would stall.
      float lse_l[4], dv_l[4];
#pragma unroll
      for (int qsub = 0; qsub < 4; ++qsub) {
        int qc = qbase + qsub * 16 + lrow;
        lse_l[qsub] = lse_b[qc];
        dv_l[qsub] = dvec_b[(long long)qc * Hq];
      }
#pragma unroll
      for (int qsub = 0; qsub < 4; ++qsub)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int qrow = qbase + qsub * 16 + lgrp * 4 + r;
          float l = __shfl(lse_l[qsub], lgrp * 4 + r, 64);
          float pv;
          if ((causal && qrow < my_kv) || l == -INFINITY)
            pv = 0.f;
          else
            pv = __expf(sp[qsub][r] * scale - l);
          sp[qsub][r] = pv;
          dp[qsub][r] = pv * (dp[qsub][r] -
                              __shfl(dv_l[qsub], lgrp * 4 + r, 64));
        }

      // P / dS -> B-fragments (same shuffle recipe as the fwd kernel).
      s16x8 p_b[2], ds_b[2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int src = (2 * (lgrp & 1) + (j >> 2)) * 16 + lrow;
          float p0 = __shfl(sp[2 * ks][j & 3], src, 64);
          float p1 = __shfl(sp[2 * ks + 1][j & 3], src, 64);
          p_b[ks][j] = (short)f2bf_trunc((lgrp >> 1) ? p1 : p0);
          float d0 = __shfl(dp[2 * ks][j & 3], src, 64);
          float d1 = __shfl(dp[2 * ks + 1][j & 3], src, 64);
          ds_b[ks][j] = (short)f2bf((lgrp >> 1) ? d1 : d0);
        }

      // dV^T += dO^T P ; dK^T += Q^T dS  (A reads identical bytes to
      // the unswapped kernel's dOT/QT B reads).
#pragma unroll
      for (int ct = 0; ct < 8; ++ct)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          int drow = ct * 16 + lrow;
          int b_off = swzT(drow * 128 + (ks * 32 + lgrp * 8) * 2, drow);
          s16x8 a_dot = *(const s16x8*)((char*)dot_lds + b_off);
          dv_t[ct] = MFMA_BF16(as_bf16x8(a_dot), as_bf16x8(p_b[ks]),
                               dv_t[ct]);
          s16x8 a_qt = *(const s16x8*)((char*)qt_lds + b_off);
          dk_t[ct] = MFMA_BF16(as_bf16x8(a_qt), as_bf16x8(ds_b[ks]),
                               dk_t[ct]);
        }
    }
  }

  // dK/dV transpose on store (s16x4 per d block), dK scaled.
  unsigned short* dKb = dK + ((long long)b * S * Hkv + kvh) * ATT_D;
  unsigned short* dVb = dV + ((long long)b * S * Hkv + kvh) * ATT_D;
  unsigned short* krow_p = dKb + (long long)my_kv * kv_rowstride;
  unsigned short* vrow_p = dVb + (long long)my_kv * kv_rowstride;
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) {
    s16x4 kv4, vv4;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      kv4[r] = (short)f2bf(dk_t[ct][r] * scale);
      vv4[r] = (short)f2bf(dv_t[ct][r]);
    }
    *(s16x4*)(krow_p + ct * 16 + lgrp * 4) = kv4;
    *(s16x4*)(vrow_p + ct * 16 + lgrp * 4) = vv4;
  }
}

// ---------------------------------------------------------------------------
// dK/dV with counted-wait transpose reads (guide §5.5 T3/T10): the
// Q^T/dO^T staging (64 scalar scatter writes/thread/tile + 32 KB LDS)
// is deleted; the dV/dK B-fragments come straight from the row-major
// q_lds/do_lds via ds_read_b64_tr_b16 in a depth-2 software pipeline —
// the i+1 fragment's four tr reads are in flight while fragment i's
// MFMAs run, with an explicit s_waitcnt lgkmcnt(4) boundary (the naive
// per-fragment lgkmcnt(0) version measured 8.6 -> 10.1 ms).  All LDS
// ops inside the pipelined region go through inline asm so the
// compiler cannot insert its own (miscounting) waits.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256, 2) void attn_bwd_dkv_tr_kernel(
    const unsigned short* __restrict__ Q, const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V, const unsigned short* __restrict__ dO,
    const float* __restrict__ lse, const float* __restrict__ Dvec,
    unsigned short* __restrict__ dK, unsigned short* __restrict__ dV, int B,
    int S, int Hq, int Hkv, float scale, int causal) {
  __shared__ unsigned short q_lds[BM * ATT_D];
  __shared__ unsigned short do_lds[BM * ATT_D];
  __shared__ unsigned short p_lds[BN * BM];   // PT
  __shared__ unsigned short ds_lds[BN * BM];  // dST

  const int kt = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hkv;
  const int kvh = bh % Hkv;
  const int group = Hq / Hkv;
  const int kvbase = kt * BN;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int lrow = lane & 15;
  const int lgrp = lane >> 4;

  const long long q_rowstride = (long long)Hq * ATT_D;
  const long long kv_rowstride = (long long)Hkv * ATT_D;
  const unsigned short* Kb = K + ((long long)b * S * Hkv + kvh) * ATT_D;
  const unsigned short* Vb = V + ((long long)b * S * Hkv + kvh) * ATT_D;

  s16x8 a_k[4], a_v[4];
  {
    const int kvrow = kvbase + 16 * w + lrow;
    const unsigned short* ksrc = Kb + (long long)kvrow * kv_rowstride;
    const unsigned short* vsrc = Vb + (long long)kvrow * kv_rowstride;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      a_k[ks] = *(const s16x8*)(ksrc + ks * 32 + lgrp * 8);
      a_v[ks] = *(const s16x8*)(vsrc + ks * 32 + lgrp * 8);
    }
  }

  f32x4 dv_acc[8], dk_acc[8];
#pragma unroll
  for (int ct = 0; ct < 8; ++ct) {
    dv_acc[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
    dk_acc[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  const unsigned int q_base_a = (unsigned int)(unsigned long long)q_lds;
  const unsigned int do_base_a = (unsigned int)(unsigned long long)do_lds;

  for (int g = 0; g < group; ++g) {
    const int qh = kvh * group + g;
    const unsigned short* Qb = Q + ((long long)b * S * Hq + qh) * ATT_D;
    const unsigned short* dOb = dO + ((long long)b * S * Hq + qh) * ATT_D;
    const float* lse_b = lse + ((long long)b * Hq + qh) * S;
    const float* dvec_b = Dvec + (long long)b * S * Hq + qh;

    const int qt0 = causal ? kvbase / BM : 0;
    for (int qt = qt0; qt < S / BM; ++qt) {
      const int qbase = qt * BM;
      __syncthreads();
      // Stage Q, dO (vector writes only — no transposed scatter).
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int idx = tid + i * 256;
        int row = idx >> 4, ch = idx & 15;
        s16x8 qv = *(const s16x8*)(Qb + (long long)(qbase + row) * q_rowstride + ch * 8);
        *(s16x8*)((char*)q_lds + swz(row * 256 + ch * 16, row)) = qv;
        s16x8 dov = *(const s16x8*)(dOb + (long long)(qbase + row) * q_rowstride + ch * 8);
        *(s16x8*)((char*)do_lds + swz(row * 256 + ch * 16, row)) = dov;
      }
      __syncthreads();

      // ST = K Q^T, dPT = V dO^T (unchanged from the classic kernel).
      f32x4 st[4], dpt[4];
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        st[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
        dpt[ct] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
          int qrow = ct * 16 + lrow;
          s16x8 bq = *(const s16x8*)((char*)q_lds +
                                     swz(qrow * 256 + (ks * 32 + lgrp * 8) * 2, qrow));
          st[ct] = MFMA_BF16(as_bf16x8(a_k[ks]), as_bf16x8(bq), st[ct]);
          s16x8 bdo = *(const s16x8*)((char*)do_lds +
                                      swz(qrow * 256 + (ks * 32 + lgrp * 8) * 2, qrow));
          dpt[ct] = MFMA_BF16(as_bf16x8(a_v[ks]), as_bf16x8(bdo), dpt[ct]);
        }
      }

      const int my_kvrow = kvbase + 16 * w + lgrp * 4;  // + r
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        int qcol = qbase + ct * 16 + lrow;
        float l = lse_b[qcol];
        float dvv = dvec_b[(long long)qcol * Hq];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float sv = st[ct][r] * scale;
          float pv;
          if ((causal && qcol < my_kvrow + r) || l == -INFINITY)
            pv = 0.f;
          else
            pv = __expf(sv - l);
          int prow = 16 * w + lgrp * 4 + r;
          int pcol = ct * 16 + lrow;
          int off = swzP(prow * 128 + pcol * 2, prow);
          *(unsigned short*)((char*)p_lds + off) = f2bf_trunc(pv);
          *(unsigned short*)((char*)ds_lds + off) =
              f2bf(pv * (dpt[ct][r] - dvv));
        }
      }
      __syncthreads();

      // A-fragments (PT/dST) via synchronous asm b128 reads so the
      // compiler tracks zero outstanding ds ops in the counted region.
      s16x8 pfrag[2], dsfrag[2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        int prow = 16 * w + lrow;
        unsigned a_off = (unsigned)(unsigned long long)(char*)p_lds +
            (unsigned)swzP(prow * 128 + (ks * 32 + lgrp * 8) * 2, prow);
        pfrag[ks] = ds_read_b128_sync(a_off);
        unsigned d_off = (unsigned)(unsigned long long)(char*)ds_lds +
            (unsigned)swzP(prow * 128 + (ks * 32 + lgrp * 8) * 2, prow);
        dsfrag[ks] = ds_read_b128_sync(d_off);
      }

      // Counted-wait pipeline over the 16 (ct, ks) fragments: issue
      // fragment i+1's four tr reads, wait until only those four remain
      // outstanding, MFMA fragment i.
      tr4 buf[2];
      int q0 = 0 * 32 + lgrp * 8 + (lrow >> 2);  // ks=0 of ct=0
      int din0 = 0 * 32 + (lrow & 3) * 8;
      {
        int ch0 = din0 >> 4;
        int b0 = q0 * 256 + ((ch0 ^ (q0 & 7)) << 4) + (din0 & 15);
        int q1 = q0 + 4;
        int b1 = q1 * 256 + ((ch0 ^ (q1 & 7)) << 4) + (din0 & 15);
        ds_tr4_issue(&buf[0], do_base_a + b0, do_base_a + b1,
                     q_base_a + b0, q_base_a + b1);
      }
#pragma unroll
      for (int i = 0; i < 16; ++i) {
        const int ct = i >> 1, ks = i & 1;
        if (i + 1 < 16) {
          const int nct = (i + 1) >> 1, nks = (i + 1) & 1;
          int nq0 = nks * 32 + lgrp * 8 + (lrow >> 2);
          int ndin = nct * 32 + (lrow & 3) * 8;
          int nch = ndin >> 4;
          int nb0 = nq0 * 256 + ((nch ^ (nq0 & 7)) << 4) + (ndin & 15);
          int nq1 = nq0 + 4;
          int nb1 = nq1 * 256 + ((nch ^ (nq1 & 7)) << 4) + (ndin & 15);
          ds_tr4_issue(&buf[(i + 1) & 1], do_base_a + nb0,
                       do_base_a + nb1, q_base_a + nb0, q_base_a + nb1);
          lgkm_wait4_bind(&buf[i & 1]);
        } else {
          lgkm_wait0_bind(&buf[i & 1]);
        }
        union { unsigned long long u[2]; s16x8 v; } dof, qf;
        dof.u[0] = buf[i & 1].d[0];
        dof.u[1] = buf[i & 1].d[1];
        qf.u[0] = buf[i & 1].d[2];
        qf.u[1] = buf[i & 1].d[3];
        dv_acc[ct] = MFMA_BF16(as_bf16x8(pfrag[ks]), as_bf16x8(dof.v),
                               dv_acc[ct]);
        dk_acc[ct] = MFMA_BF16(as_bf16x8(dsfrag[ks]), as_bf16x8(qf.v),
                               dk_acc[ct]);
      }
    }
  }

  unsigned short* dKb = dK + ((long long)b * S * Hkv + kvh) * ATT_D;
  unsigned short* dVb = dV + ((long long)b * S * Hkv + kvh) * ATT_D;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kvrow = kvbase + 16 * w + lgrp * 4 + r;
    unsigned short* krow = dKb + (long long)kvrow * kv_rowstride;
    unsigned short* vrow = dVb + (long long)kvrow * kv_rowstride;
#pragma unroll
    for (int ct = 0; ct < 8; ++ct) {
      krow[ct * 16 + lrow] = f2bf(dk_acc[ct][r] * scale);
      vrow[ct * 16 + lrow] = f2bf(dv_acc[ct][r]);
    }
  }
}
