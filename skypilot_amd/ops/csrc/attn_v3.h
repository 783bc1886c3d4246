// Shared helpers for the v3 (deep-pipelined 32x32) attention kernels.
#pragma once
#include "common.h"

typedef __attribute__((ext_vector_type(16))) float f32x16;
#define MFMA32V3(a, b, c) \
  __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0)

// 16-slot XOR swizzle for the K tile (rows are 256 B = 16 chunks): a
// wave's A-fragment read (32 rows, fixed 16 B chunk) spreads over all
// 16 slots -> 2 lanes/slot = free (guide §6 G4: 2-way is 1.02x).
__device__ __forceinline__ int swzK16(int byte_off, int row) {
  return byte_off ^ ((row & 15) << 4);
}

// V subtile layout: element index of V[kv][d] inside the
// [kv/4][d/16][4][16] tile.  tr-read gather addresses land on
// consecutive 8 B slots per 16-lane group (conflict-free); the staging
// s16x8 writes stay 16 B contiguous because d moves within one subtile
// row.
__device__ __forceinline__ int vsub(int kv, int d) {
  return (((kv >> 2) << 3) + (d >> 4)) * 64 + ((kv & 3) << 4) + (d & 15);
}

__device__ __forceinline__ unsigned cvtpk_bf16(float a, float b) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
  return r;
}

// lane <-> lane+32 exchange via permlane32_swap: pure VALU (unlike
// __shfl_xor's ds_bpermute, which is an LGKM op whose compiler-inserted
// lgkmcnt(0) would drain our in-flight tr reads mid-softmax).
__device__ __forceinline__ float xhalf32(float x) {
  union { float f; unsigned u; } c;
  c.f = x;
  auto p = __builtin_amdgcn_permlane32_swap(c.u, c.u, false, false);
  union { unsigned u; float f; } r0, r1;
  r0.u = p[0];
  r1.u = p[1];
  return (threadIdx.x & 32) ? r0.f : r1.f;
}

// Counted wait for a 2-deep tr-read pipeline: block until only the most
// recent 8 lgkm ops (the next batch's tr reads) remain outstanding, and
// bind the dependency to this batch's 8 destination registers.
__device__ __forceinline__ void lgkm_wait8_bind2(tr4* a, tr4* b) {
  asm volatile("s_waitcnt lgkmcnt(8)"
               : "+v"(a->d[0]), "+v"(a->d[1]), "+v"(a->d[2]),
                 "+v"(a->d[3]), "+v"(b->d[0]), "+v"(b->d[1]),
                 "+v"(b->d[2]), "+v"(b->d[3])
               :
               : "memory");
  __builtin_amdgcn_sched_barrier(0);
}
// Drain outstanding global_load_lds before a buffer-handoff barrier.
// LDS-DMA completion is tracked by VMcnt, and hipcc inserts only
// `s_waitcnt lgkmcnt(0)` before s_barrier (verified in r02 ISA) — so
// without this a wave can read staged K/V rows before the DMA lands.
// Intermittent small-S numerics failures (r02) were exactly this race.
__device__ __forceinline__ void vm_drain() {
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
}

__device__ __forceinline__ void lgkm_wait0_bind2(tr4* a, tr4* b) {
  asm volatile("s_waitcnt lgkmcnt(0)"
               : "+v"(a->d[0]), "+v"(a->d[1]), "+v"(a->d[2]),
                 "+v"(a->d[3]), "+v"(b->d[0]), "+v"(b->d[1]),
                 "+v"(b->d[2]), "+v"(b->d[3])
               :
               : "memory");
  __builtin_amdgcn_sched_barrier(0);
}

// Async 16B global->LDS copy: per-lane global source, wave-uniform LDS
// base + lane*16 destination (guide §5: the only supported dest form).
__device__ __forceinline__ void gload_lds16(const void* g, void* l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)g,
      (__attribute__((address_space(3))) unsigned int*)l, 16, 0, 0);
}


// P/dS C-layout -> MFMA fragment redistribution (T12): cvt_pk pairs +
// permlane32_swap assemble the 8 contraction elements (k = 8h + j for
// dest lane half h) from the C-layout registers of the two lane halves.
// V3_PACK2: source spans TWO C-tiles stv[2] (64-wide contraction, ks =
// 0..3, block n = ks>>1).  V3_PACK1: ONE C-tile (32-wide, ks = 0..1).
#define V3_PACK_CORE(pb, c0, c1, c2, c3, c4, c5, c6, c7)                  \
  {                                                                       \
    unsigned w0a = cvtpk_bf16(c0, c1);                                    \
    unsigned w1a = cvtpk_bf16(c2, c3);                                    \
    unsigned w0b = cvtpk_bf16(c4, c5);                                    \
    unsigned w1b = cvtpk_bf16(c6, c7);                                    \
    auto p0 = __builtin_amdgcn_permlane32_swap(w0a, w0b, false, false);   \
    auto p1 = __builtin_amdgcn_permlane32_swap(w1a, w1b, false, false);   \
    union { unsigned u[4]; s16x8 v; } pk;                                 \
    pk.u[0] = p0[0];                                                      \
    pk.u[1] = p1[0];                                                      \
    pk.u[2] = p0[1];                                                      \
    pk.u[3] = p1[1];                                                      \
    pb = pk.v;                                                            \
  }
#define V3_PACK2(pb, stv, ks)                                             \
  {                                                                       \
    const int e0 = ((ks) & 1) << 3;                                       \
    const int n = (ks) >> 1;                                              \
    V3_PACK_CORE(pb, stv[n][e0 + 0], stv[n][e0 + 1], stv[n][e0 + 2],      \
                 stv[n][e0 + 3], stv[n][e0 + 4], stv[n][e0 + 5],          \
                 stv[n][e0 + 6], stv[n][e0 + 7]);                         \
  }
#define V3_PACK1(pb, stv, ks)                                             \
  {                                                                       \
    const int e0 = (ks) << 3;                                             \
    V3_PACK_CORE(pb, stv[e0 + 0], stv[e0 + 1], stv[e0 + 2],               \
                 stv[e0 + 3], stv[e0 + 4], stv[e0 + 5], stv[e0 + 6],      \
                 stv[e0 + 7]);                                            \
  }
