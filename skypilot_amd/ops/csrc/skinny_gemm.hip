// Decode GEMV / skinny GEMM: Y[N,O] = X[N,I] @ W[O,I]^T, bf16 in/out,
// fp32 accumulate.  N = decode batch (1..8 after hipGraph bucketing).
//
// Rationale (measured, docs/BENCHMARKS.md): hipBLASLt moves ~2.1 TB/s
// effective on the decode-step [n,h]x[h,m] shapes, leaving ~3x of the
// 6.3 TB/s HBM roofline on the table.  The op is purely weight-
// bandwidth-bound: X is at most 8 x 14336 bf16 = 229 KB and stays L2
// resident while W (up to 1 GB for lm_head) streams through once.
// One wave owns one output row (a contiguous 2I-byte run of W), four
// rows per 256-thread block; each lane strides b128 loads across the
// row and keeps N fp32 partial sums; a wave-wide shuffle reduction
// finishes each dot product.  No LDS, no barriers.
//
// Reference parity note: the reference framework has no serving GEMV
// (it delegates serving to external images);
// this backs skypilot_amd.serve's single-stream latency path.
#include "common.h"

typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;
union b128 {
  s16x8 v;
  bf16x2 h[4];
};
// 2-way bf16 dot with f32 accumulate: one v_dot2c_f32_bf16 per 2 MACs
// (4 VALU ops per b128 chunk instead of 16 cvt+fma) — keeps n<=8
// weight-bandwidth-bound instead of VALU-bound.
__device__ __forceinline__ float dot2(s16x8 a, s16x8 b, float acc) {
  union b128 ua, ub;
  ua.v = a;
  ub.v = b;
#pragma unroll
  for (int j = 0; j < 4; ++j)
    acc = __builtin_amdgcn_fdot2_f32_bf16(ua.h[j], ub.h[j], acc, false);
  return acc;
}

// Multi-row variant for N >= 2 on large O (>= 16384 keeps the grid
// full at R=4): each wave owns R=4 consecutive output
// rows, so every X chunk loaded from L2 feeds 4 dot products — N=8's
// X re-read traffic through L2 (8x the W bytes at R=1) drops 4x, which
// is what capped the R=1 kernel at ~2.6 TB/s on the big shapes.
template <int N, int R>
__global__ __launch_bounds__(256) void skinny_gemm_multirow_kernel(
    const unsigned short* __restrict__ W,
    const unsigned short* __restrict__ X,
    unsigned short* __restrict__ Y, int I, int O) {
  int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  int o0 = (blockIdx.x * 4 + wave) * R;
  if (o0 >= O) return;
  float acc[R][N];
#pragma unroll
  for (int r = 0; r < R; ++r)
#pragma unroll
    for (int b = 0; b < N; ++b) acc[r][b] = 0.f;
  const unsigned short* wrow = W + (long long)o0 * I;
  for (int i = lane * 8; i < I; i += 512) {
    s16x8 wv[R];
#pragma unroll
    for (int r = 0; r < R; ++r)
      wv[r] = __builtin_nontemporal_load(
          (const s16x8*)(wrow + (long long)r * I + i));
#pragma unroll
    for (int b = 0; b < N; ++b) {
      s16x8 xv = *(const s16x8*)(X + (long long)b * I + i);
#pragma unroll
      for (int r = 0; r < R; ++r) acc[r][b] = dot2(wv[r], xv, acc[r][b]);
    }
  }
#pragma unroll
  for (int r = 0; r < R; ++r)
#pragma unroll
    for (int b = 0; b < N; ++b) {
      float v = wave_reduce_sum(acc[r][b]);
      if (lane == 0 && o0 + r < O)
        Y[(long long)b * O + o0 + r] = f2bf(v);
    }
}

template <int N>
__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    const unsigned short* __restrict__ W,
    const unsigned short* __restrict__ X,
    unsigned short* __restrict__ Y, int I, int O) {
  int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  int o = blockIdx.x * 4 + wave;
  if (o >= O) return;
  const unsigned short* wrow = W + (long long)o * I;
  // 4 independent b128 W loads in flight per lane (a single
  // loop-carried chain stalls on vmcnt), W nontemporal so X stays
  // L2-resident, and 4 accumulators per batch row so the dot2c
  // dependency chains interleave.
  float a0[N], a1[N], a2[N], a3[N];
#pragma unroll
  for (int b = 0; b < N; ++b) a0[b] = a1[b] = a2[b] = a3[b] = 0.f;
  int i = lane * 8;
  for (; i + 3 * 512 + 8 <= I; i += 4 * 512) {
    s16x8 w0 = __builtin_nontemporal_load((const s16x8*)(wrow + i));
    s16x8 w1 = __builtin_nontemporal_load((const s16x8*)(wrow + i + 512));
    s16x8 w2 = __builtin_nontemporal_load((const s16x8*)(wrow + i + 1024));
    s16x8 w3 = __builtin_nontemporal_load((const s16x8*)(wrow + i + 1536));
#pragma unroll
    for (int b = 0; b < N; ++b) {
      const unsigned short* xb = X + (long long)b * I + i;
      a0[b] = dot2(w0, *(const s16x8*)(xb), a0[b]);
      a1[b] = dot2(w1, *(const s16x8*)(xb + 512), a1[b]);
      a2[b] = dot2(w2, *(const s16x8*)(xb + 1024), a2[b]);
      a3[b] = dot2(w3, *(const s16x8*)(xb + 1536), a3[b]);
    }
  }
  for (; i < I; i += 512) {
    s16x8 wv = __builtin_nontemporal_load((const s16x8*)(wrow + i));
#pragma unroll
    for (int b = 0; b < N; ++b)
      a0[b] = dot2(wv, *(const s16x8*)(X + (long long)b * I + i), a0[b]);
  }
#pragma unroll
  for (int b = 0; b < N; ++b) {
    float r = wave_reduce_sum(a0[b] + a1[b] + a2[b] + a3[b]);
    if (lane == 0) Y[(long long)b * O + o] = f2bf(r);
  }
}

// SwiGLU-fused variant for the decode down-projection: X is the packed
// gate|up GEMV output [N, 2M]; each lane computes silu(g)*u on the fly
// (g = X[:, i], u = X[:, i+M]) so the separate swiglu kernel launch
// (one per layer per decode step) disappears.  N=1 path only — larger
// N routes to hipBLASLt where the separate swiglu amortizes.
__device__ __forceinline__ float _silu(float g) {
  return g / (1.f + __expf(-g));
}

template <int N>
__global__ __launch_bounds__(256) void skinny_gemm_swiglu_kernel(
    const unsigned short* __restrict__ W,   // [O, M]
    const unsigned short* __restrict__ X,   // [N, 2M] packed gate|up
    unsigned short* __restrict__ Y, int M, int O) {
  int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  int o = blockIdx.x * 4 + wave;
  if (o >= O) return;
  const unsigned short* wrow = W + (long long)o * M;
  float a0[N], a1[N], a2[N], a3[N];
#pragma unroll
  for (int b = 0; b < N; ++b) a0[b] = a1[b] = a2[b] = a3[b] = 0.f;
  // same 4-deep load pipeline as the plain GEMV (a single chain stalls
  // on vmcnt and cost -15% end-to-end when measured naively)
  int i = lane * 8;
  for (; i + 3 * 512 + 8 <= M; i += 4 * 512) {
    s16x8 w0 = __builtin_nontemporal_load((const s16x8*)(wrow + i));
    s16x8 w1 = __builtin_nontemporal_load((const s16x8*)(wrow + i + 512));
    s16x8 w2 = __builtin_nontemporal_load((const s16x8*)(wrow + i + 1024));
    s16x8 w3 = __builtin_nontemporal_load((const s16x8*)(wrow + i + 1536));
#pragma unroll
    for (int b = 0; b < N; ++b) {
      const unsigned short* gb = X + (long long)b * 2 * M + i;
      const unsigned short* ub = gb + M;
      s16x8 g0 = *(const s16x8*)(gb);
      s16x8 g1 = *(const s16x8*)(gb + 512);
      s16x8 g2 = *(const s16x8*)(gb + 1024);
      s16x8 g3 = *(const s16x8*)(gb + 1536);
      s16x8 u0 = *(const s16x8*)(ub);
      s16x8 u1 = *(const s16x8*)(ub + 512);
      s16x8 u2 = *(const s16x8*)(ub + 1024);
      s16x8 u3 = *(const s16x8*)(ub + 1536);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        a0[b] += bf2f(w0[j]) * (_silu(bf2f((unsigned short)g0[j])) *
                                bf2f((unsigned short)u0[j]));
        a1[b] += bf2f(w1[j]) * (_silu(bf2f((unsigned short)g1[j])) *
                                bf2f((unsigned short)u1[j]));
        a2[b] += bf2f(w2[j]) * (_silu(bf2f((unsigned short)g2[j])) *
                                bf2f((unsigned short)u2[j]));
        a3[b] += bf2f(w3[j]) * (_silu(bf2f((unsigned short)g3[j])) *
                                bf2f((unsigned short)u3[j]));
      }
    }
  }
  float acc[N];
#pragma unroll
  for (int b = 0; b < N; ++b)
    acc[b] = a0[b] + a1[b] + a2[b] + a3[b];
  for (; i < M; i += 512) {
    s16x8 wv = __builtin_nontemporal_load((const s16x8*)(wrow + i));
#pragma unroll
    for (int b = 0; b < N; ++b) {
      const unsigned short* xb = X + (long long)b * 2 * M;
      s16x8 gv = *(const s16x8*)(xb + i);
      s16x8 uv = *(const s16x8*)(xb + M + i);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        acc[b] += bf2f(wv[j]) *
                  (_silu(bf2f((unsigned short)gv[j])) *
                   bf2f((unsigned short)uv[j]));
    }
  }
#pragma unroll
  for (int b = 0; b < N; ++b) {
    float r = wave_reduce_sum(acc[b]);
    if (lane == 0) Y[(long long)b * O + o] = f2bf(r);
  }
}

extern "C" void skinny_gemm_swiglu_launch(const void* W, const void* X,
                                          void* Y, int N, int M, int O,
                                          hipStream_t stream) {
  dim3 grid((O + 3) / 4), block(256);
#define SCASE(n)                                                          case n:                                                                   hipLaunchKernelGGL(skinny_gemm_swiglu_kernel<n>, grid, block, 0,                           stream, (const unsigned short*)W,                                       (const unsigned short*)X, (unsigned short*)Y, M,                        O);                                                  break;
  switch (N) {
    SCASE(1) SCASE(2) SCASE(3) SCASE(4)
    default:
      break;
  }
#undef SCASE
}

extern "C" void skinny_gemm_launch(const void* W, const void* X, void* Y,
                                   int N, int I, int O,
                                   hipStream_t stream) {
  dim3 grid((O + 3) / 4), block(256);
  const unsigned short* w = (const unsigned short*)W;
  const unsigned short* x = (const unsigned short*)X;
  unsigned short* y = (unsigned short*)Y;
  dim3 grid_mr((O + 15) / 16);
#define CASE(n)                                                        \
  case n:                                                              \
    if (n >= 2 && O >= 16384 && O % 4 == 0)                            \
      hipLaunchKernelGGL((skinny_gemm_multirow_kernel<n, 4>), grid_mr, \
                         block, 0, stream, w, x, y, I, O);             \
    else                                                               \
      hipLaunchKernelGGL(skinny_gemm_kernel<n>, grid, block, 0,        \
                         stream, w, x, y, I, O);                       \
    break;
  switch (N) {
    CASE(1) CASE(2) CASE(3) CASE(4) CASE(5) CASE(6) CASE(7) CASE(8)
    default:
      break;  // host wrapper guarantees 1 <= N <= 8
  }
#undef CASE
}

// ---------------------------------------------------------------------------
// fp8 (OCP e4m3fn) weight GEMV: Y[N,O] = X[N,I] @ (scale[o] * W8[O,I])^T.
// Decode is weight-BANDWIDTH-bound (header comment): storing W as fp8
// with one fp32 scale per output row halves the streamed bytes, so the
// single-stream decode rate doubles at the same 6.5-7.3 TB/s roofline.
// Hardware fp8->f32 conversion (v_cvt_pk_f32_fp8, 2 elems/op) keeps the
// kernel bandwidth-bound; X stays bf16 (exact), accumulation fp32.
// ---------------------------------------------------------------------------
union fp8x16 {
  i32x4 i;
  unsigned int u[4];
};

__device__ __forceinline__ void dot_fp8_16(unsigned int w4x4[4],
                                           const unsigned short* xb,
                                           float* acc) {
  float xf[16];
#pragma unroll
  for (int j = 0; j < 16; ++j) xf[j] = bf2f(xb[j]);
#pragma unroll
  for (int q = 0; q < 4; ++q) {
    f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(w4x4[q], false);
    f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(w4x4[q], true);
    *acc += lo[0] * xf[q * 4 + 0] + lo[1] * xf[q * 4 + 1] +
            hi[0] * xf[q * 4 + 2] + hi[1] * xf[q * 4 + 3];
  }
}

template <int N>
__global__ __launch_bounds__(256) void skinny_gemm_fp8_kernel(
    const unsigned char* __restrict__ W8,
    const float* __restrict__ scale,
    const unsigned short* __restrict__ X,
    unsigned short* __restrict__ Y, int I, int O) {
  int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  int o = blockIdx.x * 4 + wave;
  if (o >= O) return;
  const unsigned char* wrow = W8 + (long long)o * I;
  // 4 independent 16B loads in flight (same pipeline shape as the bf16
  // kernel: a single loop-carried chain stalls on vmcnt).
  float a0[N], a1[N], a2[N], a3[N];
#pragma unroll
  for (int b = 0; b < N; ++b) a0[b] = a1[b] = a2[b] = a3[b] = 0.f;
  int i = lane * 16;
  for (; i + 3 * 1024 + 16 <= I; i += 4 * 1024) {
    fp8x16 w0, w1, w2, w3;
    w0.i = __builtin_nontemporal_load((const i32x4*)(wrow + i));
    w1.i = __builtin_nontemporal_load((const i32x4*)(wrow + i + 1024));
    w2.i = __builtin_nontemporal_load((const i32x4*)(wrow + i + 2048));
    w3.i = __builtin_nontemporal_load((const i32x4*)(wrow + i + 3072));
#pragma unroll
    for (int b = 0; b < N; ++b) {
      const unsigned short* xb = X + (long long)b * I + i;
      dot_fp8_16(w0.u, xb, &a0[b]);
      dot_fp8_16(w1.u, xb + 1024, &a1[b]);
      dot_fp8_16(w2.u, xb + 2048, &a2[b]);
      dot_fp8_16(w3.u, xb + 3072, &a3[b]);
    }
  }
  for (; i < I; i += 1024) {
    fp8x16 wv;
    wv.i = __builtin_nontemporal_load((const i32x4*)(wrow + i));
#pragma unroll
    for (int b = 0; b < N; ++b)
      dot_fp8_16(wv.u, X + (long long)b * I + i, &a0[b]);
  }
#pragma unroll
  for (int b = 0; b < N; ++b) {
    float r = wave_reduce_sum(a0[b] + a1[b] + a2[b] + a3[b]);
    if (lane == 0) Y[(long long)b * O + o] = f2bf(r * scale[o]);
  }
}

// Multi-row fp8 variant: R rows per wave share one X conversion and
// keep 2R 16B loads in flight (the R=1 kernel is latency-bound at fp8
// row sizes — half the bytes of bf16 per row).
template <int N, int R>
__global__ __launch_bounds__(256) void skinny_gemm_fp8_mr_kernel(
    const unsigned char* __restrict__ W8,
    const float* __restrict__ scale,
    const unsigned short* __restrict__ X,
    unsigned short* __restrict__ Y, int I, int O) {
  int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  int o0 = (blockIdx.x * 4 + wave) * R;
  if (o0 >= O) return;
  const unsigned char* wrow = W8 + (long long)o0 * I;
  float acc[R][N];
#pragma unroll
  for (int r = 0; r < R; ++r)
#pragma unroll
    for (int b = 0; b < N; ++b) acc[r][b] = 0.f;
  int i = lane * 16;
  for (; i + 1024 + 16 <= I; i += 2048) {
    fp8x16 w[R][2];
#pragma unroll
    for (int r = 0; r < R; ++r) {
      w[r][0] = *(fp8x16*)&(((const i32x4*)(wrow + (long long)r * I + i))[0]);
      w[r][1] = *(fp8x16*)&(((const i32x4*)(wrow + (long long)r * I + i + 1024))[0]);
    }
#pragma unroll
    for (int b = 0; b < N; ++b) {
      const unsigned short* xb = X + (long long)b * I + i;
#pragma unroll
      for (int r = 0; r < R; ++r) {
        dot_fp8_16(w[r][0].u, xb, &acc[r][b]);
        dot_fp8_16(w[r][1].u, xb + 1024, &acc[r][b]);
      }
    }
  }
  for (; i < I; i += 1024) {
    fp8x16 w[R];
#pragma unroll
    for (int r = 0; r < R; ++r)
      w[r].i = __builtin_nontemporal_load(
          (const i32x4*)(wrow + (long long)r * I + i));
#pragma unroll
    for (int b = 0; b < N; ++b)
#pragma unroll
      for (int r = 0; r < R; ++r)
        dot_fp8_16(w[r].u, X + (long long)b * I + i, &acc[r][b]);
  }
#pragma unroll
  for (int r = 0; r < R; ++r)
#pragma unroll
    for (int b = 0; b < N; ++b) {
      float v = wave_reduce_sum(acc[r][b]);
      if (lane == 0 && o0 + r < O)
        Y[(long long)b * O + o0 + r] = f2bf(v * scale[o0 + r]);
    }
}

extern "C" void skinny_gemm_fp8_launch(const void* W8, const float* scale,
                                       const void* X, void* Y, int N,
                                       int I, int O, hipStream_t stream) {
  // SKY_FP8_MR: 0 = single-row kernel, else rows-per-wave (default 2).
  static const int mr = [] {
    const char* e = getenv("SKY_FP8_MR");
    return e ? atoi(e) : 2;
  }();
  if (mr >= 2 && O % 2 == 0) {
    dim3 gridm((O / 2 + 3) / 4), block(256);
#define F8MR(n)                                                          case n:                                                                  hipLaunchKernelGGL((skinny_gemm_fp8_mr_kernel<n, 2>), gridm,                              block, 0, stream, (const unsigned char*)W8,                            scale, (const unsigned short*)X,                                       (unsigned short*)Y, I, O);                          break;
    switch (N) {
      F8MR(1) F8MR(2) F8MR(3) F8MR(4) F8MR(5) F8MR(6) F8MR(7) F8MR(8)
      default:
        break;
    }
#undef F8MR
    return;
  }
  dim3 grid((O + 3) / 4), block(256);
#define F8CASE(n)                                                      \
  case n:                                                              \
    hipLaunchKernelGGL(skinny_gemm_fp8_kernel<n>, grid, block, 0,      \
                       stream, (const unsigned char*)W8, scale,        \
                       (const unsigned short*)X, (unsigned short*)Y,   \
                       I, O);                                          \
    break;
  switch (N) {
    F8CASE(1) F8CASE(2) F8CASE(3) F8CASE(4)
    F8CASE(5) F8CASE(6) F8CASE(7) F8CASE(8)
    default:
      break;
  }
#undef F8CASE
}

// Norm-fused fp8 GEMV: rmsnorm(x [+ res]) * nw computed ONCE per block
// into LDS, then the standard multi-row weight-streaming loops read
// the normed input from LDS.  Deletes the separate rmsnorm_res launch
// before the qkv / gate_up GEMVs — decode is dispatch-gap-bound (~5 us
// idle PER KERNEL inside graph replay, profiles/
// r02_fp8_decode_kernel_stats.txt), so one fewer launch saves kernel
// time AND gap.  Unlike the silu-in-GEMV negative result (per-output-
// row recompute), the LDS staging makes the norm a per-BLOCK prologue
// that all blocks run in parallel (~2 us wall).  Block 0 additionally
// writes x+res back for the residual stream.  I <= 4096 (h of the
// norm-fed projections; N <= 2 fp8 rows fit 16 KB LDS).
template <int N, int R>
__global__ __launch_bounds__(256) void skinny_gemm_fp8_norm_kernel(
    const unsigned char* __restrict__ W8, const float* __restrict__ scale,
    const unsigned short* __restrict__ X,    // [N, I] raw x
    const unsigned short* __restrict__ RES,  // [N, I] residual or null
    const unsigned short* __restrict__ NW,   // [I] norm weight
    unsigned short* __restrict__ XOUT,       // [N, I] x+res or null
    float eps, unsigned short* __restrict__ Y, int I, int O) {
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  __shared__ unsigned short xn[N][4096];
  __shared__ float rred[4];
  // ---- prologue: t = x (+ res); rms over f32 t; xn = t*inv*nw ------
#pragma unroll
  for (int b = 0; b < N; ++b) {
    float ss = 0.f;
    for (int j = threadIdx.x; j < I; j += 256) {
      float t = bf2f(X[(long long)b * I + j]);
      if (RES) t += bf2f(RES[(long long)b * I + j]);
      xn[b][j] = f2bf(t);
      ss += t * t;
      if (XOUT && blockIdx.x == 0)
        XOUT[(long long)b * I + j] = f2bf(t);
    }
    ss = wave_reduce_sum(ss);
    if (lane == 0) rred[wave] = ss;
    __syncthreads();
    float inv = rsqrtf((rred[0] + rred[1] + rred[2] + rred[3]) / I + eps);
    __syncthreads();  // rred reused next b; xn[b] writes drained
    for (int j = threadIdx.x; j < I; j += 256) {
      float t = bf2f(xn[b][j]);
      xn[b][j] = f2bf(t * inv * bf2f(NW[j]));
    }
  }
  __syncthreads();
  // ---- standard multi-row fp8 weight stream over LDS-resident xn ---
  int o0 = (blockIdx.x * 4 + wave) * R;
  if (o0 >= O) return;
  const unsigned char* wrow = W8 + (long long)o0 * I;
  float acc[R][N];
#pragma unroll
  for (int r = 0; r < R; ++r)
#pragma unroll
    for (int b = 0; b < N; ++b) acc[r][b] = 0.f;
  int i = lane * 16;
  for (; i + 1024 + 16 <= I; i += 2048) {
    fp8x16 w[R][2];
#pragma unroll
    for (int r = 0; r < R; ++r) {
      w[r][0] = *(fp8x16*)&(((const i32x4*)(wrow + (long long)r * I + i))[0]);
      w[r][1] =
          *(fp8x16*)&(((const i32x4*)(wrow + (long long)r * I + i + 1024))[0]);
    }
#pragma unroll
    for (int b = 0; b < N; ++b) {
      const unsigned short* xb = &xn[b][i];
#pragma unroll
      for (int r = 0; r < R; ++r) {
        dot_fp8_16(w[r][0].u, xb, &acc[r][b]);
        dot_fp8_16(w[r][1].u, xb + 1024, &acc[r][b]);
      }
    }
  }
  for (; i < I; i += 1024) {
    fp8x16 w[R];
#pragma unroll
    for (int r = 0; r < R; ++r)
      w[r].i = __builtin_nontemporal_load(
          (const i32x4*)(wrow + (long long)r * I + i));
#pragma unroll
    for (int b = 0; b < N; ++b)
#pragma unroll
      for (int r = 0; r < R; ++r) dot_fp8_16(w[r].u, &xn[b][i], &acc[r][b]);
  }
#pragma unroll
  for (int r = 0; r < R; ++r)
#pragma unroll
    for (int b = 0; b < N; ++b) {
      float v = wave_reduce_sum(acc[r][b]);
      if (lane == 0 && o0 + r < O)
        Y[(long long)b * O + o0 + r] = f2bf(v * scale[o0 + r]);
    }
}

extern "C" void skinny_gemm_fp8_norm_launch(
    const void* W8, const float* scale, const void* X, const void* RES,
    const void* NW, void* XOUT, float eps, void* Y, int N, int I, int O,
    hipStream_t stream) {
  dim3 gridm((O / 2 + 3) / 4), block(256);
#define F8NORM(n)                                                        \
  case n:                                                                \
    hipLaunchKernelGGL((skinny_gemm_fp8_norm_kernel<n, 2>), gridm,       \
                       block, 0, stream, (const unsigned char*)W8,       \
                       scale, (const unsigned short*)X,                  \
                       (const unsigned short*)RES,                       \
                       (const unsigned short*)NW,                        \
                       (unsigned short*)XOUT, eps, (unsigned short*)Y,   \
                       I, O);                                            \
    break;
  switch (N) {
    F8NORM(1) F8NORM(2)
    default:
      break;
  }
#undef F8NORM
}
