// Weight-gradient GEMM path: transpose-then-NT (CDNA4/gfx950).
//
// wgrad is a TN GEMM (dW[i,j] = sum_m dy[m,i] * x[m,j]): BOTH operands
// have the contraction axis strided, which defeats the fast staged-LDS
// fragment reads.  hipBLASLt floors at ~1,000 TF/s on these shapes
// (round-1 measurement, docs/BENCHMARKS.md).  Instead we transpose both
// operands once (memory-bound tiled kernel, ~2x tensor bytes of
// traffic) and run an NT GEMM (C[i,j] = sum_m A[i,m] * B[j,m], both
// m-contiguous) built on the guide's 256x256 8-phase schedule
// (cdna_hip_programming.md §5.5 T1-T5: raw s_barrier phases, counted
// waits, global_load_lds staging with pre-swizzled sources, XOR LDS
// swizzle, setprio around MFMA clusters).
//
// No reference counterpart (SkyPilot ships no kernels).
#include <cstdlib>

#include "common.h"

typedef __attribute__((ext_vector_type(16))) float f32x16;
#define MFMA32W(a, b, c) \
  __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0)

// ---------------------------------------------------------------------------
// Tiled bf16 transpose: out[c][r] = in[r][c].  64x64 tiles through
// padded LDS; vectorized s16x8 on both sides of the staging.
// ---------------------------------------------------------------------------
#define TP 72  // padded LDS row (elements)
extern "C" __global__ __launch_bounds__(256) void transpose_bf16_kernel(
    const unsigned short* __restrict__ in, unsigned short* __restrict__ out,
    int R, int C) {
  __shared__ unsigned short t[64 * TP];
  const int tiles_r = R >> 6;
  const int tiles_c = C >> 6;
  const int tid = threadIdx.x;
  for (long long tile = blockIdx.x; tile < (long long)tiles_r * tiles_c;
       tile += gridDim.x) {
    const int tr = (int)(tile / tiles_c);
    const int tc = (int)(tile % tiles_c);
    const long long r0 = (long long)tr << 6;
    const long long c0 = (long long)tc << 6;
    __syncthreads();
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int r = (tid >> 3) + 32 * i;
      int c8 = (tid & 7) * 8;
      *(s16x8*)&t[r * TP + c8] =
          *(const s16x8*)(in + (r0 + r) * C + c0 + c8);
    }
    __syncthreads();
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int c = (tid >> 3) + 32 * i;
      int r8 = (tid & 7) * 8;
      s16x8 v;
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = t[(r8 + j) * TP + c];
      *(s16x8*)(out + (c0 + c) * R + r0 + r8) = v;
    }
  }
}

// global_load_lds helper local to this TU (same as attn_v3.h's).
__device__ __forceinline__ void gload_lds16w(const void* g, void* l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)g,
      (__attribute__((address_space(3))) unsigned int*)l, 16, 0, 0);
}

// ---------------------------------------------------------------------------
// NT GEMM, 8-phase schedule.  C[i,j] = sum_m A[i,m] * B[j,m].
//   A [I, M] row-major, B [J, M] row-major, C [I, J] bf16 out.
//   Block tile BM x 256 (BM = 256 or 128 for small-I shapes), BK = 64.
//   8 waves as 2x4: wave tile (BM/2) x 64.
//   LDS: A [BM][64] + B [256][64] bf16, XOR-swizzled rows (8 chunk
//   slots), double-buffered; staged by global_load_lds with the
//   inverse swizzle applied to the SOURCE address (linear dest).
//   Phases: one per k-quarter (ks = 0..3): {frag ds_reads, one
//   half-tile prefetch issue, s_barrier, lgkmcnt(0), 8*BM/256 MFMA
//   cluster in setprio(1), s_barrier}; vmcnt(0) once per K-tile.
// ---------------------------------------------------------------------------
template <int BM>
__global__ __launch_bounds__(512, 1) void gemm_nt_kernel(
    const unsigned short* __restrict__ A, const unsigned short* __restrict__ B,
    unsigned short* __restrict__ C, int I, int J, long long M,
    int swz_grid) {
  // LDS: [BM][64] A-tile + [256][64] B-tile, x2 buffers.
  __shared__ unsigned short a_lds[2][BM * 64];
  __shared__ unsigned short b_lds[2][256 * 64];

  constexpr int MT = BM / 64;  // i-tiles per wave (4 or 2)
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int wm = w >> 2;  // 0..1
  const int wn = w & 3;   // 0..3
  const int col = lane & 31;
  const int h = lane >> 5;

  // XCD-bijective block swizzle (guide m204): consecutive logical tiles
  // land on the same XCD's L2.
  int bid = blockIdx.x;
  if (swz_grid) {
    const int nwg = gridDim.x;
    const int q = nwg >> 3, r = nwg & 7;
    const int xcd = bid & 7, off = bid >> 3;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
  }
  const int nbj = J >> 8;
  const int bi = bid / nbj;
  const int bj = bid % nbj;
  const long long i0 = (long long)bi * BM;
  const long long j0 = (long long)bj << 8;

  f32x16 acc[MT][2];
#pragma unroll
  for (int it = 0; it < MT; ++it)
#pragma unroll
    for (int jt = 0; jt < 2; ++jt)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[it][jt][r] = 0.f;

  // ---- staging plan.  Each gl_lds covers 8 rows x 8 chunks (1 KB).
  // A: BM*128B = BM/8 insts; B: 32 insts; per wave: BM/64 + 4.
  const int srow = lane >> 3;          // 0..7 within an inst
  const int schk = lane & 7;           // 16B chunk
#define WG_STAGE_A(bufi, kt)                                              \
  {                                                                       \
    _Pragma("unroll") for (int ii = 0; ii < BM / 64; ++ii) {              \
      int r0 = (w * (BM / 64) + ii) * 8;                                  \
      int row = r0 + srow;                                                \
      gload_lds16w(                                                       \
          A + (i0 + row) * M + (long long)(kt) * 64 +                     \
              ((schk ^ (row & 7)) << 3),                                  \
          (char*)a_lds[bufi] + r0 * 128);                                 \
    }                                                                     \
  }
#define WG_STAGE_B(bufi, kt)                                              \
  {                                                                       \
    _Pragma("unroll") for (int ii = 0; ii < 4; ++ii) {                    \
      int r0 = (w * 4 + ii) * 8;                                          \
      int row = r0 + srow;                                                \
      gload_lds16w(                                                       \
          B + (j0 + row) * M + (long long)(kt) * 64 +                     \
              ((schk ^ (row & 7)) << 3),                                  \
          (char*)b_lds[bufi] + r0 * 128);                                 \
    }                                                                     \
  }

  const int n_kt = (int)(M >> 6);
  WG_STAGE_A(0, 0);
  WG_STAGE_B(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int kt = 0; kt < n_kt; ++kt) {
    const int buf = kt & 1;
    const bool more = kt + 1 < n_kt;
    // 4 phases, one per k-quarter.  Frags: MT a + 2 b ds_read_b128.
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      s16x8 af[MT], bf[2];
      const int chk = (2 * ks + h) << 4;  // byte offset of the 16B chunk
#pragma unroll
      for (int it = 0; it < MT; ++it) {
        int row = wm * (BM / 2) + it * 32 + col;
        af[it] = *(const s16x8*)(
            (char*)a_lds[buf] + row * 128 + (chk ^ ((row & 7) << 4)));
      }
#pragma unroll
      for (int jt = 0; jt < 2; ++jt) {
        int row = wn * 64 + jt * 32 + col;
        bf[jt] = *(const s16x8*)(
            (char*)b_lds[buf] + row * 128 + (chk ^ ((row & 7) << 4)));
      }
      // staged prefetch: one half-tile per phase (A halves then B).
      if (more) {
        if (ks == 0) WG_STAGE_A(buf ^ 1, kt + 1);
        if (ks == 1) WG_STAGE_B(buf ^ 1, kt + 1);
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int it = 0; it < MT; ++it)
#pragma unroll
        for (int jt = 0; jt < 2; ++jt)
          acc[it][jt] =
              MFMA32W(as_bf16x8(af[it]), as_bf16x8(bf[jt]), acc[it][jt]);
      __builtin_amdgcn_s_setprio(0);
      if (ks == 3) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }

  // ---- epilogue: bf16 store; lane's j is contiguous within a tile.
#pragma unroll
  for (int it = 0; it < MT; ++it) {
    const long long ib = i0 + wm * (BM / 2) + it * 32;
#pragma unroll
    for (int jt = 0; jt < 2; ++jt) {
      const long long jb = j0 + wn * 64 + jt * 32;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        long long irow = ib + (r & 3) + ((r >> 2) << 3) + (h << 2);
        C[irow * J + jb + col] = f2bf(acc[it][jt][r]);
      }
    }
  }
}

extern "C" void transpose_bf16_launch(const void* in, void* out, int R,
                                      int C, hipStream_t stream) {
  long long tiles = ((long long)R >> 6) * (C >> 6);
  int grid = tiles > 4096 ? 4096 : (int)tiles;
  hipLaunchKernelGGL(transpose_bf16_kernel, dim3(grid), dim3(256), 0,
                     stream, (const unsigned short*)in,
                     (unsigned short*)out, R, C);
}

extern "C" void gemm_nt_launch(const void* A, const void* B, void* C, int I,
                               int J, long long M, hipStream_t stream) {
  // BM=128 when I is small (fills the CUs: 512-thread blocks, 1/CU).
  const long long blocks256 = ((long long)I >> 8) * (J >> 8);
  if (I % 256 == 0 && blocks256 >= 512) {
    dim3 grid((unsigned)blocks256);
    hipLaunchKernelGGL(gemm_nt_kernel<256>, grid, dim3(512), 0, stream,
                       (const unsigned short*)A, (const unsigned short*)B,
                       (unsigned short*)C, I, J, M, 1);
  } else {
    dim3 grid((unsigned)(((long long)I >> 7) * (J >> 8)));
    hipLaunchKernelGGL(gemm_nt_kernel<128>, grid, dim3(512), 0, stream,
                       (const unsigned short*)A, (const unsigned short*)B,
                       (unsigned short*)C, I, J, M, 1);
  }
}
