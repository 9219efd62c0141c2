// Common device utilities for the gfx950 (CDNA4, MI355X) kernels.
// Pure HIP — no torch headers, no CUDA compatibility shims.
//
// Conventions (see /root/repo/SURVEY.md §2d for the op inventory):
//  - wave = 64 lanes (CDNA), block sizes are multiples of 64
//  - activations are [N,H,W,C] with C innermost; C is a multiple of 64 in
//    every conv/BN layer (ResNet: 64..2048), so channel vectors load as
//    16-byte ushort8 (bf16/fp16) or float4 chunks
//  - compute in fp32, storage dtype templated over float/bf16/f16
//  - memory-bound kernels: grid = min(work, 2048 blocks) + grid-stride
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define PDTA_DT_F32 0
#define PDTA_DT_BF16 1
#define PDTA_DT_F16 2

#define DEV_INLINE __device__ __forceinline__

constexpr int kWave = 64;

// ---- scalar dtype conversion --------------------------------------------
DEV_INLINE float to_f32(float x) { return x; }
DEV_INLINE float to_f32(__hip_bfloat16 x) { return __bfloat162float(x); }
DEV_INLINE float to_f32(__half x) { return __half2float(x); }

template <typename T> DEV_INLINE T from_f32(float x);
template <> DEV_INLINE float from_f32<float>(float x) { return x; }
template <> DEV_INLINE __hip_bfloat16 from_f32<__hip_bfloat16>(float x) {
  return __float2bfloat16(x);
}
template <> DEV_INLINE __half from_f32<__half>(float x) {
  return __float2half(x);
}

// ---- 8-wide value vector (the 16-B/lane coalescing sweet spot for 16-bit
// dtypes; 32 B for fp32 done as two float4) ------------------------------
template <typename T> struct Vec8 { T v[8]; };

template <typename T>
DEV_INLINE Vec8<T> load8(const T* p) {
  Vec8<T> r;
  if constexpr (sizeof(T) == 2) {
    // one 16-byte load
    *reinterpret_cast<uint4*>(r.v) = *reinterpret_cast<const uint4*>(p);
  } else {
    *reinterpret_cast<uint4*>(&r.v[0]) = *reinterpret_cast<const uint4*>(p);
    *reinterpret_cast<uint4*>(&r.v[4]) = *reinterpret_cast<const uint4*>(p + 4);
  }
  return r;
}

template <typename T>
DEV_INLINE void store8(T* p, const Vec8<T>& r) {
  if constexpr (sizeof(T) == 2) {
    *reinterpret_cast<uint4*>(p) = *reinterpret_cast<const uint4*>(r.v);
  } else {
    *reinterpret_cast<uint4*>(p) = *reinterpret_cast<const uint4*>(&r.v[0]);
    *reinterpret_cast<uint4*>(p + 4) =
        *reinterpret_cast<const uint4*>(&r.v[4]);
  }
}

// ---- wave reductions (64-wide) ------------------------------------------
DEV_INLINE float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, 64);
  return x;
}

DEV_INLINE float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, 64));
  return x;
}

// ---- division by launch-invariant divisor (magic numbers; validated
// exhaustively for n < 2^31 over the divisors this framework uses) -------
struct MagicDiv {
  unsigned mul;
  int shift;
};
static inline MagicDiv magic_div(unsigned d) {
  MagicDiv m;
  if ((d & (d - 1)) == 0) {  // power of two: ceil(2^s/d) = 2^32 truncates
    int k = 0;
    while ((1u << k) < d) ++k;
    m.mul = 0x80000000u;  // n >> k == (n * 2^31) >> (31 + k)
    m.shift = 31 + k;
    return m;
  }
  // non-power-of-two: shift = 31 + ceil(log2 d) keeps the multiplier in
  // 32 bits (a 32+L shift makes it 33 bits, which silently truncated);
  // valid for n < 2^31 — all index spaces in this framework.
  int s = 0;
  while ((1u << s) < d) ++s;
  m.shift = 31 + s;
  m.mul = (unsigned)(((1ull << m.shift) + d - 1) / d);
  return m;
}
DEV_INLINE unsigned mdiv(unsigned n, unsigned mul, int shift) {
  return (unsigned)(((unsigned long long)n * mul) >> shift);
}

// ---- grid sizing ---------------------------------------------------------
static inline int ew_grid(long n, int tpb, int per_thread) {
  long blocks = (n + (long)tpb * per_thread - 1) / ((long)tpb * per_thread);
  if (blocks > 2048) blocks = 2048;  // grid-stride the rest (G11)
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

// 16-bit dtypes only (MFMA kernels: bf16/f16; fp32 compute is a CPU-config
// concern, not a GPU path)
#define PDTA_DISPATCH_DTYPE16(DT, TYPE, ...)        \
  switch (DT) {                                     \
    case PDTA_DT_BF16: {                            \
      using TYPE = __hip_bfloat16;                  \
      __VA_ARGS__;                                  \
      break;                                        \
    }                                               \
    case PDTA_DT_F16: {                             \
      using TYPE = __half;                          \
      __VA_ARGS__;                                  \
      break;                                        \
    }                                               \
  }

#define PDTA_DISPATCH_DTYPE(DT, TYPE, ...)          \
  switch (DT) {                                     \
    case PDTA_DT_F32: {                             \
      using TYPE = float;                           \
      __VA_ARGS__;                                  \
      break;                                        \
    }                                               \
    case PDTA_DT_BF16: {                            \
      using TYPE = __hip_bfloat16;                  \
      __VA_ARGS__;                                  \
      break;                                        \
    }                                               \
    case PDTA_DT_F16: {                             \
      using TYPE = __half;                          \
      __VA_ARGS__;                                  \
      break;                                        \
    }                                               \
  }
