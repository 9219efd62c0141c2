// Shared MFMA tile machinery for the GEMM-shaped kernels (gemm.hip,
// conv.hip): fragment types, 16x16x32 MFMA dispatch, swizzled LDS
// addressing, global_load_lds staging, XCD-aware block swizzle.
#pragma once
#include "common.h"

using bf16 = __hip_bfloat16;
using f16 = __half;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short s16x8;

template <typename T> struct MfmaOp;
template <> struct MfmaOp<bf16> {
  static DEV_INLINE f32x4 run(s16x8 a, s16x8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  }
};
template <> struct MfmaOp<f16> {
  static DEV_INLINE f32x4 run(s16x8 a, s16x8 b, f32x4 c) {
    typedef __attribute__((ext_vector_type(8))) _Float16 f16x8;
    union U { s16x8 s; f16x8 h; };
    U ua, ub;
    ua.s = a;
    ub.s = b;
    return __builtin_amdgcn_mfma_f32_16x16x32_f16(ua.h, ub.h, c, 0, 0, 0);
  }
};

// LDS operand tile [rows][64 k-elements] of 2-B dtype: 128-B rows with the
// bank-conflict XOR swizzle on the 16-B chunk index (T2; write side is the
// pre-swizzled glds SOURCE address — guide §5.4 rule 21).
DEV_INLINE int lds_off128(int row, int col16) {
  return row * 128 + (col16 ^ (row & 7)) * 16;
}

DEV_INLINE s16x8 frag_read128(const char* lds, int row, int col16) {
  return *reinterpret_cast<const s16x8*>(lds + lds_off128(row, col16));
}

// one glds wave-instruction: 64 lanes x 16 B -> 1 KiB at a wave-uniform
// LDS base (lane-linear); per-lane global source address.
DEV_INLINE void glds16(const void* gsrc, void* lds_dst) {
  typedef const __attribute__((address_space(1))) uint32_t* gp_t;
  typedef __attribute__((address_space(3))) uint32_t* lp_t;
  __builtin_amdgcn_global_load_lds((gp_t)gsrc, (lp_t)lds_dst, 16, 0, 0);
}

// Transposed fragment read via ds_read_b64_tr_b16 (gfx950 hardware
// transpose; mapping probed on silicon — gpurun_out/tr_probe.log): a
// 16-lane group's concatenated 8-B fetches (in lane order) form a [4][16]
// row-major 16-bit matrix and each lane receives its (lane&15) column.
// Reads an MFMA A/B fragment (lane l: element j = M[m0 + (l&15)]
// [k0 + (l>>4)*8 + j]) from a [k][m] LDS image with 128-B rows carrying
// the chunk^=(k&7) 16-B swizzle (write side pre-swizzles the glds source).
// Issue form: the two tr reads are ISSUED without a wait so many
// fragments' reads fly concurrently; the caller places ONE
// tr_read_wait() before the first consumer (a per-call lgkmcnt(0) was a
// serial ~50-cycle chain per fragment pair — 47% issue-stall in the wrw
// kernel's PMC). Outputs are early-clobber and only valid after the wait.
DEV_INLINE void frag_read_tr_issue(unsigned lds_base, int k0, int m0,
                                   int lane, unsigned long long* lo,
                                   unsigned long long* hi) {
  int g = lane >> 4;
  int row = (lane & 15) >> 2;  // which of the 4 k-rows this lane fetches
  int seg = lane & 3;          // which 8-B segment of the 32-B m-span
  int byte0 = m0 * 2 + seg * 8;
  int chunk = byte0 >> 4, rem = byte0 & 15;
  int ka = k0 + g * 8 + row;      // rows for elements j=0..3
  int kb = ka + 4;                // rows for elements j=4..7
  unsigned a0 = lds_base + ka * 128 + (((chunk ^ (ka & 7)) << 4) | rem);
  unsigned a1 = lds_base + kb * 128 + (((chunk ^ (kb & 7)) << 4) | rem);
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3"
      : "=&v"(*lo), "=&v"(*hi)
      : "v"(a0), "v"(a1));
}

// Offset-precompute split of frag_read_tr_issue: the (k0, m0, lane)
// address math is loop-invariant in ring-buffered K-loops — compute the
// in-slot byte offsets ONCE, then each tile's read is base+off (2 VALU)
// instead of the full ~10-op recompute per fragment pair.
DEV_INLINE void tr_off(int k0, int m0, int lane, unsigned* off0,
                       unsigned* off1) {
  int g = lane >> 4;
  int row = (lane & 15) >> 2;
  int seg = lane & 3;
  int byte0 = m0 * 2 + seg * 8;
  int chunk = byte0 >> 4, rem = byte0 & 15;
  int ka = k0 + g * 8 + row;
  int kb = ka + 4;
  *off0 = ka * 128 + (((chunk ^ (ka & 7)) << 4) | rem);
  *off1 = kb * 128 + (((chunk ^ (kb & 7)) << 4) | rem);
}

// ROW_BYTES variant of tr_off (full-chunk-range XOR, matching
// frag_read_tr_issue_s<ROW_BYTES>)
template <int ROW_BYTES>
DEV_INLINE void tr_off_s(int k0, int m0, int lane, unsigned* off0,
                         unsigned* off1) {
  constexpr int KMASK = ROW_BYTES / 16 - 1;
  int g = lane >> 4;
  int row = (lane & 15) >> 2;
  int seg = lane & 3;
  int byte0 = m0 * 2 + seg * 8;
  int chunk = byte0 >> 4, rem = byte0 & 15;
  int ka = k0 + g * 8 + row;
  int kb = ka + 4;
  *off0 = ka * ROW_BYTES + (((chunk ^ (ka & KMASK)) << 4) | rem);
  *off1 = kb * ROW_BYTES + (((chunk ^ (kb & KMASK)) << 4) | rem);
}

DEV_INLINE void tr_issue_at(unsigned a0, unsigned a1,
                            unsigned long long* lo,
                            unsigned long long* hi) {
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3"
      : "=&v"(*lo), "=&v"(*hi)
      : "v"(a0), "v"(a1));
}

// Stride-parameterized variant: [k][m] LDS image with ROW_BYTES-wide rows
// (e.g. 512 B for 256-feature tiles). The XOR swizzle uses the FULL chunk
// range of the row (k & (ROW_BYTES/16 - 1)): one ds_read_b64_tr_b16
// touches rows up to 24 apart across its four 16-lane groups, and a 3-bit
// mask left rows 8/16/24 apart on identical banks (PMC: 18% of wrw-big
// wave cycles in LDS bank-conflict stalls; 6 conflict cycles per access
// vs the 2-cycle floor of an 8-B/lane read).
template <int ROW_BYTES>
DEV_INLINE void frag_read_tr_issue_s(unsigned lds_base, int k0, int m0,
                                     int lane, unsigned long long* lo,
                                     unsigned long long* hi) {
  constexpr int KMASK = ROW_BYTES / 16 - 1;
  int g = lane >> 4;
  int row = (lane & 15) >> 2;
  int seg = lane & 3;
  int byte0 = m0 * 2 + seg * 8;
  int chunk = byte0 >> 4, rem = byte0 & 15;
  int ka = k0 + g * 8 + row;
  int kb = ka + 4;
  unsigned a0 =
      lds_base + ka * ROW_BYTES + (((chunk ^ (ka & KMASK)) << 4) | rem);
  unsigned a1 =
      lds_base + kb * ROW_BYTES + (((chunk ^ (kb & KMASK)) << 4) | rem);
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3"
      : "=&v"(*lo), "=&v"(*hi)
      : "v"(a0), "v"(a1));
}

// Row-pitch-parameterized plain fragment read (frag_read128 for images
// whose rows are ROW_BYTES wide; same 16-B-chunk XOR swizzle).
template <int ROW_BYTES>
DEV_INLINE s16x8 frag_read_s(const char* lds, int row, int chunk) {
  return *reinterpret_cast<const s16x8*>(
      lds + row * ROW_BYTES + ((chunk ^ (row & 7)) * 16));
}

// Wait for every outstanding LDS read, then fence the scheduler so no
// consumer (incl. register-only MFMA — guide §5.4 rule 18) is hoisted
// above the wait.
DEV_INLINE void tr_read_wait() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
}

DEV_INLINE s16x8 tr_pair_to_frag(unsigned long long lo,
                                 unsigned long long hi) {
  union {
    struct { unsigned long long lo, hi; } u;
    s16x8 v;
  } cvt;
  cvt.u.lo = lo;
  cvt.u.hi = hi;
  return cvt.v;
}

// LDS byte offset of a generic pointer to __shared__ memory
DEV_INLINE unsigned lds_addr_of(const void* p) {
  return (unsigned)(uintptr_t)(__attribute__((address_space(3))) const void*)
      p;
}

// bijective XCD swizzle (T1): contiguous chunk of the grid per XCD
DEV_INLINE int xcd_swizzle(int bid, int nwg) {
  int q = nwg / 8, r = nwg % 8;
  int xcd = bid % 8, idx = bid / 8;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}
