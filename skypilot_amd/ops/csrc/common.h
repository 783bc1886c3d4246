// Common device helpers for skypilot_amd CDNA4 (gfx950) kernels.
//
// All kernels in this tree are written directly for MI355X: wave64,
// 32-bank LDS, MFMA 16x16x32 bf16 tiles. No CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#define WAVE 64

// ---------------------------------------------------------------------------
// Vector types (ext_vector_type so hipcc maps them onto VGPR tuples).
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(2))) float  f32x2;
typedef __attribute__((ext_vector_type(4))) float  f32x4;
typedef __attribute__((ext_vector_type(8))) float  f32x8;
typedef __attribute__((ext_vector_type(4))) short  s16x4;
typedef __attribute__((ext_vector_type(8))) short  s16x8;
typedef __attribute__((ext_vector_type(2))) short  s16x2;
typedef __attribute__((ext_vector_type(4))) int    i32x4;
typedef __attribute__((ext_vector_type(2))) int    i32x2;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

union bf8_cast {
  s16x8 s;
  bf16x8 b;
};
__device__ __forceinline__ bf16x8 as_bf16x8(s16x8 v) {
  bf8_cast c;
  c.s = v;
  return c.b;
}

// MFMA 16x16x32 bf16 fragment layouts on gfx950 (verified on-device by
// tests/test_gpu_mfma.py probe):
//   A[16x32]: lane l holds A[l&15][(l>>4)*8 + j]        j=0..7
//   B[32x16]: lane l holds B[(l>>4)*8 + j][l&15]
//   C/D[16x16]: lane l reg r holds C[(l>>4)*4 + r][l&15]
#define MFMA_BF16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0)

// XOR swizzle for row-major LDS tiles with 128B/256B row stride: spreads a
// column slice across 8 16B slots so wave-wide ds_read_b128 at fixed column
// is ~conflict-free (guide §6 Guideline 4).
__device__ __forceinline__ int swz(int byte_off, int row) {
  return byte_off ^ ((row & 7) << 4);
}

// Swizzle for TRANSPOSED tiles ([d][seq] layouts): their scatter-writes
// walk d in strides of 8 (one 16B source chunk covers d = 8c..8c+7), so
// (row&7) is constant per write instruction — keying on it left every
// write a 32-way bank conflict (measured: ~1-2e9 SQ_LDS_BANK_CONFLICT
// per attention dispatch).  Key on (row>>3) instead: varies per lane on
// the write side, spans 2 values per 16-row read slice on the read side
// (reads stay balanced: the chunk index itself varies with lane group).
__device__ __forceinline__ int swzT(int byte_off, int row) {
  return byte_off ^ (((row >> 3) & 7) << 4);
}

// Read-friendly transposed-tile swizzle: keyed on the LOW d bits so the
// B/A-fragment reads (lanes sweep d = ct*16 + lrow, identical kv
// offset) spread across banks (swzT keyed d>>3 leaves them ~4-way
// conflicted — the dominant LDS stall measured in attention).  Writes
// must then be the vectorized transpose-staging path (s16x4 at d-pairs
// spanning ch and quad), which this keying also spreads.
__device__ __forceinline__ int swzT2(int byte_off, int row) {
  return byte_off ^ ((row & 7) << 4);
}

// Swizzle for the P / dS staging tiles ([q or kv rows][64] C-layout
// scatter): writes vary the row as lgrp*4+r (only 2 distinct (row&7)
// per instruction) but cols span 16 — keying on (row>>1) gives 4 row
// classes x 8 col-words = all 32 banks on the write side while the
// 16-consecutive-row fragment reads stay balanced.
__device__ __forceinline__ int swzP(int byte_off, int row) {
  return byte_off ^ (((row >> 1) & 7) << 4);
}

// Paired ds_read_b64_tr_b16: two transpose-reads + one waitcnt.
// Semantics (HW-verified by trb16 probe, 2026-09-12): per 16-lane
// group, out[lw][j] = in_lane[(lw>>2)+4j][lw&3] where in_lane[r] is the
// 4 bf16 at lane r's address.  With per-lane addresses
//   q(r) = qbase + (r>>2),  d(r) = dbase + (r&3)*4
// over a row-major [seq][D] tile, the two reads assemble exactly the
// MFMA B-fragment B[k=q][n=d] — no transposed staging copy needed.
// The sched_barrier is required (guide rule #18): hipcc hoists
// register-only MFMA past inline-asm lgkmcnt.
__device__ __forceinline__ s16x8 ds_tr_b16_pair(unsigned int a0,
                                                unsigned int a1) {
  unsigned long long lo, hi;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=v"(lo), "=v"(hi)
      : "v"(a0), "v"(a1)
      : "memory");
  __builtin_amdgcn_sched_barrier(0);
  union { unsigned long long u[2]; s16x8 v; } cvt;
  cvt.u[0] = lo;
  cvt.u[1] = hi;
  return cvt.v;
}

// Split-phase tr reads for counted-wait pipelines (guide §5.5 T3):
// ds_tr4_issue fires FOUR ds_read_b64_tr_b16 without waiting;
// lgkm_wait4_bind blocks until only the most recent 4 lgkm ops remain
// outstanding and binds the dependency to the destination registers so
// the compiler cannot hoist consumers above the wait.  All LDS traffic
// in the pipelined region must go through these helpers (a stray
// compiler ds op breaks the count).
struct tr4 {
  unsigned long long d[4];
};
__device__ __forceinline__ void ds_tr4_issue(tr4* t, unsigned a0,
                                             unsigned a1, unsigned a2,
                                             unsigned a3) {
  asm volatile(
      "ds_read_b64_tr_b16 %0, %4\n\t"
      "ds_read_b64_tr_b16 %1, %5\n\t"
      "ds_read_b64_tr_b16 %2, %6\n\t"
      "ds_read_b64_tr_b16 %3, %7"
      : "=v"(t->d[0]), "=v"(t->d[1]), "=v"(t->d[2]), "=v"(t->d[3])
      : "v"(a0), "v"(a1), "v"(a2), "v"(a3)
      : "memory");
}
__device__ __forceinline__ void lgkm_wait4_bind(tr4* t) {
  asm volatile("s_waitcnt lgkmcnt(4)"
               : "+v"(t->d[0]), "+v"(t->d[1]), "+v"(t->d[2]),
                 "+v"(t->d[3])
               :
               : "memory");
  __builtin_amdgcn_sched_barrier(0);
}
__device__ __forceinline__ void lgkm_wait0_bind(tr4* t) {
  asm volatile("s_waitcnt lgkmcnt(0)"
               : "+v"(t->d[0]), "+v"(t->d[1]), "+v"(t->d[2]),
                 "+v"(t->d[3])
               :
               : "memory");
  __builtin_amdgcn_sched_barrier(0);
}

// Synchronous b128 LDS read through inline asm: used before a
// counted-wait region so NO compiler-tracked ds op remains outstanding
// (the compiler would otherwise insert its own s_waitcnt mid-region and
// count our in-flight tr reads).
__device__ __forceinline__ s16x8 ds_read_b128_sync(unsigned int addr) {
  f32x4 d;
  asm volatile("ds_read_b128 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(d)
               : "v"(addr)
               : "memory");
  __builtin_amdgcn_sched_barrier(0);
  union { f32x4 f; s16x8 s; } u;
  u.f = d;
  return u.s;
}

// bf16 <-> f32 via bit ops (we deliberately avoid __hip_bfloat16 so these
// headers stay independent of HIP half/bf16 operator macros).
__device__ __forceinline__ float bf2f(unsigned short h) {
  union { unsigned int u; float f; } v;
  v.u = ((unsigned int)h) << 16;
  return v.f;
}

__device__ __forceinline__ unsigned short f2bf(float f) {
  union { float f; unsigned int u; } v;
  v.f = f;
  unsigned int u = v.u;
  // NaN -> canonical bf16 NaN
  if ((u & 0x7fffffffu) > 0x7f800000u) return 0x7fc0;
  // round-to-nearest-even
  unsigned int round = 0x7fffu + ((u >> 16) & 1u);
  return (unsigned short)((u + round) >> 16);
}

// Convert 8 bf16 (as s16x8) to 8 floats.
__device__ __forceinline__ void bf8_to_f32(const s16x8 v, float* out) {
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = bf2f((unsigned short)v[i]);
}

// Truncating f32->bf16 (no rounding): for values already carrying
// >=bf16 rounding error (attention probabilities), saves ~4 VALU ops
// per element on the hot softmax->LDS path.
__device__ __forceinline__ unsigned short f2bf_trunc(float f) {
  union { float f; unsigned int u; } v;
  v.f = f;
  return (unsigned short)(v.u >> 16);
}

__device__ __forceinline__ s16x8 f32_to_bf8(const float* in) {
  s16x8 v;
#pragma unroll
  for (int i = 0; i < 8; ++i) v[i] = (short)f2bf(in[i]);
  return v;
}

// ---------------------------------------------------------------------------
// Wave-level reductions (64-wide).
// ---------------------------------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// Block-level reduce for blocks of up to 1024 threads (multiple of 64).
// Returns the reduced value on every thread. `lds` must hold >= 16 floats.
__device__ __forceinline__ float block_reduce_sum(float x, float* lds) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nwaves = blockDim.x / WAVE;
  x = wave_reduce_sum(x);
  if (lane == 0) lds[wid] = x;
  __syncthreads();
  float r = (lane < nwaves) ? lds[lane] : 0.f;
  r = wave_reduce_sum(r);  // cheap: only first nwaves lanes nonzero
  __syncthreads();
  return r;
}

__device__ __forceinline__ float block_reduce_max(float x, float* lds) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nwaves = blockDim.x / WAVE;
  x = wave_reduce_max(x);
  if (lane == 0) lds[wid] = x;
  __syncthreads();
  float r = (lane < nwaves) ? lds[lane] : -INFINITY;
  r = wave_reduce_max(r);
  __syncthreads();
  return r;
}

// ---------------------------------------------------------------------------
// Grid sizing helper: memory-bound kernels cap the grid and grid-stride.
// 256 CUs x 8 blocks/CU (see MI355X guide, Guideline 11).
// ---------------------------------------------------------------------------
#define MAX_MEMBOUND_BLOCKS 2048

__host__ __forceinline__ int membound_grid(long long work_items, int block) {
  long long g = (work_items + block - 1) / block;
  if (g > MAX_MEMBOUND_BLOCKS) g = MAX_MEMBOUND_BLOCKS;
  if (g < 1) g = 1;
  return (int)g;
}

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,    \
             __LINE__);                                                     \
    }                                                                       \
  } while (0)
