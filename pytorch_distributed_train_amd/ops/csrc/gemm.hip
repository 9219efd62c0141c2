// MFMA GEMM for gfx950 (SURVEY.md §2d FC/GEMM row; the "transformer MFMA
// GEMM path" of BASELINE config #5 starts here).
//
// Shape convention: "nt" — C[M,N] = A[M,K] · B[N,K]^T with BOTH operands
// row-major and K contiguous (torch Linear weight is already [N,K]; the
// "nn"/"tn" variants are expressed as nt over cheaply-packed transposed
// operands — see ops/functional.py). K must be a multiple of 64 (python
// pads); M and N tails are clamp-loaded and mask-stored.
//
// Structure (cdna_hip_programming.md §5, the "minimum 2-phase" schedule):
//   - 128x128 output tile, BK=64, 4 waves (256 threads), 64x64 per wave,
//     4x4 fragments of v_mfma_f32_16x16x32_{bf16,f16}, fp32 accumulate
//   - global->LDS staging via __builtin_amdgcn_global_load_lds width 16
//     (the compiler never auto-emits it), double-buffered
//   - LDS XOR swizzle col16 ^= (row&7) applied on the *source* address
//     (glds writes lane-linear — guide §5.4 rule 21) and on fragment reads
//   - XCD-aware bijective blockIdx swizzle (T1) for L2 locality
#include <cstdlib>

#include "common.h"
#include "mfma_tile.h"

namespace {

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int TPB = 256;  // 4 waves, 2x2 wave grid of 64x64 sub-tiles

// stage one 128x64 operand tile: 4 waves x 4 glds, each glds = 64 lanes x
// 16 B = 8 rows of 128 B, lane-linear dest; source address carries the
// swizzle (rule 21); rows clamp to row_max (tails masked at the store).
template <typename T>
DEV_INLINE void stage_tile(const T* __restrict__ src, long ld,
                           int row0, int row_max, char* lds_base, int wave,
                           int lane) {
  int row_in_chunk = lane >> 3;
  int col16 = lane & 7;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row = wave * 32 + i * 8 + row_in_chunk;
    int gcol16 = col16 ^ (row & 7);
    int grow = row0 + row;
    if (grow > row_max) grow = row_max;
    const char* gptr = reinterpret_cast<const char*>(src) +
                       ((long)grow * ld) * 2 + gcol16 * 16;
    glds16(gptr, lds_base + (wave * 32 + i * 8) * 128);
  }
}


template <typename T, typename OutT, bool BIAS>
__global__ __launch_bounds__(TPB) void gemm_nt_k(
    const T* __restrict__ A, const T* __restrict__ B, OutT* __restrict__ C,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n,
    long batch_sa, long batch_sb, long batch_sc) {
  // blockIdx.y = batch index (strided batched GEMM; attention uses
  // batch = B*heads with per-head strides)
  A += (long)blockIdx.y * batch_sa;
  B += (long)blockIdx.y * batch_sb;
  C += (long)blockIdx.y * batch_sc;
  __shared__ __attribute__((aligned(16))) char lds[2][2][BM * BK * 2];

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 1, wn = wave & 1;  // 2x2 wave grid

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  int m0 = tm * BM, n0 = tn * BN;
  int ktiles = K / BK;

  // bias values hoisted so the loads fly under the K-loop (loading
  // bias[col] inside the store loop was latency-exposed and redundant)
  float badd[4] = {0.f, 0.f, 0.f, 0.f};
  if (BIAS) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wn * 64 + j * 16 + (lane & 15);
      badd[j] = bias[col < N ? col : N - 1];
      // pin NOW: otherwise the compiler sinks these loads into the
      // epilogue and every store cluster leads with s_waitcnt vmcnt(0),
      // which (vmcnt counts stores too) serializes the whole store
      // stream — measured +28% on the small-K ViT shapes
      asm volatile("" : "+v"(badd[j]));
    }
  }

  stage_tile<T>(A, K, m0, M - 1, lds[0][0], wave, lane);
  stage_tile<T>(B, K, n0, N - 1, lds[0][1], wave, lane);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int kt = 0; kt < ktiles; ++kt) {
    if (kt + 1 < ktiles) {
      stage_tile<T>(A + (long)(kt + 1) * BK, K, m0, M - 1, lds[cur ^ 1][0],
                    wave, lane);
      stage_tile<T>(B + (long)(kt + 1) * BK, K, n0, N - 1, lds[cur ^ 1][1],
                    wave, lane);
    }
    const char* la = lds[cur][0];
    const char* lb = lds[cur][1];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // two K=32 steps per BK=64
      s16x8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int row = wm * 64 + i * 16 + (lane & 15);
        af[i] = frag_read128(la, row, (lane >> 4) + kk * 4);
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = wn * 64 + j * 16 + (lane & 15);
        bf[j] = frag_read128(lb, row, (lane >> 4) + kk * 4);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = MfmaOp<T>::run(af[i], bf[j], acc[i][j]);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: acc[i][j] reg r -> C[m0+wm*64+i*16+(lane>>4)*4+r]
  //                              [n0+wn*64+j*16+(lane&15)]
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int rbase = m0 + wm * 64 + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wn * 64 + j * 16 + (lane & 15);
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = rbase + r;
        if (row < M)
          C[(long)row * N + col] = from_f32<OutT>(acc[i][j][r] + badd[j]);
      }
    }
  }
}


// ---- large-M GEMM: 256x256 tile, 8 waves, 2-slot K-ring ----------------
// Big-tile pipelined core: per 512-thread block a 256x256 output tile
// (wave grid 2x4 -> 128x64 per wave, 64 MFMA/K-tile/wave), two 64-KiB
// K-slots (A 256x64 + B 256x64), tile T+1's eight glds issued at the top
// of tile T's compute so the ~L2 latency hides under ~2200 cycles of
// MFMA; one raw s_barrier + vmcnt drain per K-tile. Halves the staged
// bytes per FLOP vs the 128-wide tiles (the glds path, not HBM, is the
// bound at these shapes).
template <typename T, typename OutT, bool BIAS>
__global__ __launch_bounds__(512) void gemm_nt_big_k(
    const T* __restrict__ A, const T* __restrict__ B, OutT* __restrict__ C,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n,
    long batch_sa, long batch_sb, long batch_sc) {
  constexpr int BOFF = 256 * 64 * 2;  // A image bytes per slot
  __shared__ __attribute__((aligned(16))) char lds[2][2 * BOFF];

  A += (long)blockIdx.y * batch_sa;
  B += (long)blockIdx.y * batch_sb;
  C += (long)blockIdx.y * batch_sc;

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 2, wn = wave & 3;  // 2 x 4
  int m0 = tm * 256, n0 = tn * 256;
  int ktiles = K / 64;

  int row_in_chunk = lane >> 3;
  int col16 = lane & 7;

  auto stage_all = [&](int kt, int slot) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = (wave * 4 + i) * 8 + row_in_chunk;
      int grow = m0 + row;
      if (grow > M - 1) grow = M - 1;
      const char* g = reinterpret_cast<const char*>(A) +
                      ((long)grow * K + kt * 64) * 2 +
                      (col16 ^ (row & 7)) * 16;
      glds16(g, (char*)lds[slot] + (wave * 4 + i) * 8 * 128);
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = (wave * 4 + i) * 8 + row_in_chunk;
      int grow = n0 + row;
      if (grow > N - 1) grow = N - 1;
      const char* g = reinterpret_cast<const char*>(B) +
                      ((long)grow * K + kt * 64) * 2 +
                      (col16 ^ (row & 7)) * 16;
      glds16(g, (char*)lds[slot] + BOFF + (wave * 4 + i) * 8 * 128);
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // bias hoisted pre-K-loop: the in-store-loop bias[col] load was 32
  // latency-exposed scalar loads per lane (8x redundant) — measured
  // +26-139% on the small-K ViT shapes (gemm_probe PDTA_PROBE_BIAS)
  float badd[4] = {0.f, 0.f, 0.f, 0.f};
  if (BIAS) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wn * 64 + j * 16 + (lane & 15);
      badd[j] = bias[col < N ? col : N - 1];
      // pin NOW: otherwise the compiler sinks these loads into the
      // epilogue and every store cluster leads with s_waitcnt vmcnt(0),
      // which (vmcnt counts stores too) serializes the whole store
      // stream — measured +28% on the small-K ViT shapes
      asm volatile("" : "+v"(badd[j]));
    }
  }

  stage_all(0, 0);

  for (int kt = 0; kt < ktiles; ++kt) {
    int slot = kt & 1;
    // drain this tile's loads (they had the previous tile's ~2200-cycle
    // MFMA stream to fly) and rendezvous
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    // issue the NEXT tile's staging first so it spans this tile's compute
    if (kt + 1 < ktiles) stage_all(kt + 1, slot ^ 1);
    const char* la = lds[slot];
    const char* lb = lds[slot] + BOFF;
    s16x8 bf[2][4];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bf[kk][j] = frag_read128(lb, wn * 64 + j * 16 + (lane & 15),
                                 (lane >> 4) + kk * 4);
#pragma unroll
    for (int q = 0; q < 4; ++q) {  // quadrants of the wave's 128 rows
      s16x8 af[2][2];
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int i = 0; i < 2; ++i)
          af[kk][i] = frag_read128(
              la, wm * 128 + q * 32 + i * 16 + (lane & 15),
              (lane >> 4) + kk * 4);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int i = 0; i < 2; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[q * 2 + i][j] =
                MfmaOp<T>::run(af[kk][i], bf[kk][j], acc[q * 2 + i][j]);
    }
    // all waves done reading this slot before it is re-staged next tile
    __builtin_amdgcn_s_barrier();
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
    long rbase = m0 + wm * 128 + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wn * 64 + j * 16 + (lane & 15);
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = rbase + r;
        if (row < M)
          C[(long)row * N + col] = from_f32<OutT>(acc[i][j][r] + badd[j]);
      }
    }
  }
}


// ---- 8-phase 256x256 schedule (guide "8-phase template" geometry):
// same tile/wave/fragment layout as gemm_nt_big_k, but the K-loop is
// split into 4 phases per K-tile (one C-quadrant each), with per-phase
// raw-barrier pairs, s_setprio(1) around the MFMA burst, and the next
// tile's staging spread one 64-row batch per phase. The whole-tile
// rendezvous of the 2-phase structure (52% SQ_WAIT at 2 waves/SIMD) is
// replaced by 8 short lockstep phases that keep ds_read/glds/MFMA
// interleaved — the documented lever past the ~900 TF plain-HIP ceiling.
template <typename T, typename OutT, bool BIAS>
__global__ __launch_bounds__(512) void gemm_nt_8p_k(
    const T* __restrict__ A, const T* __restrict__ B, OutT* __restrict__ C,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n,
    long batch_sa, long batch_sb, long batch_sc) {
  constexpr int BOFF = 256 * 64 * 2;  // 32 KiB per operand per slot
  __shared__ __attribute__((aligned(16))) char lds[2][2 * BOFF];

  A += (long)blockIdx.y * batch_sa;
  B += (long)blockIdx.y * batch_sb;
  C += (long)blockIdx.y * batch_sc;

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 2, wn = wave & 3;  // 2 x 4
  int m0 = tm * 256, n0 = tn * 256;
  int ktiles = K / 64;

  int row_in_chunk = lane >> 3;
  int col16 = lane & 7;

  // one 64-row batch (8 KiB; one glds16 per thread) of operand A (b<4)
  // or B (b>=4) for K-tile kt into `slot`
  auto stage_batch = [&](int kt, int slot, int b) {
    int i = b & 3;
    int row = (wave * 4 + i) * 8 + row_in_chunk;
    if (b < 4) {
      int grow = m0 + row;
      if (grow > M - 1) grow = M - 1;
      const char* g = reinterpret_cast<const char*>(A) +
                      ((long)grow * K + kt * 64) * 2 +
                      (col16 ^ (row & 7)) * 16;
      glds16(g, (char*)lds[slot] + (wave * 4 + i) * 8 * 128);
    } else {
      int grow = n0 + row;
      if (grow > N - 1) grow = N - 1;
      const char* g = reinterpret_cast<const char*>(B) +
                      ((long)grow * K + kt * 64) * 2 +
                      (col16 ^ (row & 7)) * 16;
      glds16(g, (char*)lds[slot] + BOFF + (wave * 4 + i) * 8 * 128);
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

#pragma unroll
  for (int b = 0; b < 8; ++b) stage_batch(0, 0, b);

  for (int kt = 0; kt < ktiles; ++kt) {
    int slot = kt & 1;
    // publish this slot: drain own glds, then whole-block rendezvous
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    const char* la = lds[slot];
    const char* lb = lds[slot] + BOFF;
    s16x8 bf[2][4];
#pragma unroll
    for (int q = 0; q < 4; ++q) {  // one phase per C-quadrant
      if (q == 0) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            bf[kk][j] = frag_read128(lb, wn * 64 + j * 16 + (lane & 15),
                                     (lane >> 4) + kk * 4);
      }
      s16x8 af[2][2];
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int i = 0; i < 2; ++i)
          af[kk][i] = frag_read128(
              la, wm * 128 + q * 32 + i * 16 + (lane & 15),
              (lane >> 4) + kk * 4);
      if (kt + 1 < ktiles) {  // spread next tile's staging over phases
        stage_batch(kt + 1, slot ^ 1, q);
        stage_batch(kt + 1, slot ^ 1, q + 4);
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int i = 0; i < 2; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[q * 2 + i][j] =
                MfmaOp<T>::run(af[kk][i], bf[kk][j], acc[q * 2 + i][j]);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
    long rbase = m0 + wm * 128 + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wn * 64 + j * 16 + (lane & 15);
      if (col >= N) continue;
      float badd = BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = rbase + r;
        if (row < M)
          C[(long)row * N + col] = from_f32<OutT>(acc[i][j][r] + badd);
      }
    }
  }
}


// Rows are 64 B (32 K-elems); swizzle XOR uses the row's 4 chunks.
DEV_INLINE s16x8 frag_read64(const char* lds, int row, int chunk) {
  return *reinterpret_cast<const s16x8*>(
      lds + row * 64 + ((chunk ^ (row & 3)) * 16));
}

// ---- 4-slot BK=32 ring: 256x256 tile, 8 waves, staging
// runs THREE K-tiles ahead so the boundary wait is a counted
// s_waitcnt vmcnt(8) — loads stay in flight across every barrier, never
// drained to 0 (the documented ~20%% stall of the 2-slot schedule).
template <typename T, typename OutT, bool BIAS>
__global__ __launch_bounds__(512) void gemm_nt_r4_k(
    const T* __restrict__ A, const T* __restrict__ B, OutT* __restrict__ C,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n,
    long batch_sa, long batch_sb, long batch_sc) {
  constexpr int BOFF = 256 * 32 * 2;  // 16 KiB per operand per slot
  __shared__ __attribute__((aligned(16))) char lds[4][2 * BOFF];

  A += (long)blockIdx.y * batch_sa;
  B += (long)blockIdx.y * batch_sb;
  C += (long)blockIdx.y * batch_sc;

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 2, wn = wave & 3;  // 2 x 4
  int m0 = tm * 256, n0 = tn * 256;
  int ktiles = K / 32;

  int row_in = lane >> 2, c4 = lane & 3;  // 64-B rows: 4 chunks of 16 B

  // one K32 tile of A+B: 2 glds per thread (A: 16 KiB, B: 16 KiB)
  auto stage_all = [&](int kt, int slot) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int row = (wave * 2 + i) * 16 + row_in;
      int grow = m0 + row;
      if (grow > M - 1) grow = M - 1;
      const char* g = reinterpret_cast<const char*>(A) +
                      ((long)grow * K + kt * 32) * 2 +
                      ((c4 ^ (row_in & 3)) * 16);
      glds16(g, (char*)lds[slot] + (wave * 2 + i) * 1024);
    }
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int row = (wave * 2 + i) * 16 + row_in;
      int grow = n0 + row;
      if (grow > N - 1) grow = N - 1;
      const char* g = reinterpret_cast<const char*>(B) +
                      ((long)grow * K + kt * 32) * 2 +
                      ((c4 ^ (row_in & 3)) * 16);
      glds16(g, (char*)lds[slot] + BOFF + (wave * 2 + i) * 1024);
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  stage_all(0, 0);
  if (ktiles > 1) stage_all(1, 1);
  if (ktiles > 2) stage_all(2, 2);

  for (int kt = 0; kt < ktiles; ++kt) {
    int slot = kt & 3;
    // slots kt+1, kt+2 may stay in flight: 2 slots x 4 glds/thread = 8
    if (kt + 2 < ktiles)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else if (kt + 1 < ktiles)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (kt + 3 < ktiles) stage_all(kt + 3, (kt + 3) & 3);
    const char* la = lds[slot];
    const char* lb = lds[slot] + BOFF;
    s16x8 bf[4];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      bf[j] = frag_read64(lb, wn * 64 + j * 16 + (lane & 15), lane >> 4);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      s16x8 af[2];
#pragma unroll
      for (int i = 0; i < 2; ++i)
        af[i] = frag_read64(la, wm * 128 + q * 32 + i * 16 + (lane & 15),
                            lane >> 4);
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[q * 2 + i][j] =
              MfmaOp<T>::run(af[i], bf[j], acc[q * 2 + i][j]);
    }
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
    long rbase = m0 + wm * 128 + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wn * 64 + j * 16 + (lane & 15);
      if (col >= N) continue;
      float badd = BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = rbase + r;
        if (row < M)
          C[(long)row * N + col] = from_f32<OutT>(acc[i][j][r] + badd);
      }
    }
  }
}

// column sum: db[n] = sum_m dy[m][n]  (linear bias grad)
template <typename T>
__global__ void colsum_f32_k(const T* __restrict__ dy, float* __restrict__ db,
                             long M, int N) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= N) return;
  long r0 = blockIdx.y, rs = gridDim.y;
  float acc = 0.f;
  for (long r = r0; r < M; r += rs) acc += to_f32(dy[r * N + c]);
  atomicAdd(&db[c], acc);
}

// ---- v2 experimental: 256x128 tile, BK=32, 8 waves (4x2) ---------------

// ---- half-tile pipelined 256x256 (PDTA_GEMM_8P2) — EXPERIMENTAL, DO
// NOT ENABLE IN PRODUCTION. Measured: no speedup over the 2-slot
// schedule (per-phase barrier cost cancels the counted-vmcnt win) AND
// an intermittent numerics deviation at K=4096 (suspected staging
// lifetime race not yet localized). Kept as the starting point for the
// next schedule investigation; see docs/DESIGN.md §5/§6.
// Design: halves (128 rows) of each operand stage into per-half slots
// with staggered lifetimes:
//   - B-halves of tile t+2 stage at t's q1/q2 (B fragments are
//     register-resident after each tile's q0, so B slots free early);
//   - A-halves of tile t+1 stage at t's q0 (A slots free after q3).
// The tile-boundary wait is then s_waitcnt vmcnt(4) — never 0.
template <typename T, typename OutT, bool BIAS>
__global__ __launch_bounds__(512) void gemm_nt_8p2_k(
    const T* __restrict__ A, const T* __restrict__ B, OutT* __restrict__ C,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n,
    long batch_sa, long batch_sb, long batch_sc) {
  constexpr int HB = 128 * 64 * 2;  // 16 KiB per half
  // [slot][A0,A1,B0,B1]
  __shared__ __attribute__((aligned(16))) char lds[2][4][HB];

  A += (long)blockIdx.y * batch_sa;
  B += (long)blockIdx.y * batch_sb;
  C += (long)blockIdx.y * batch_sc;

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 2, wn = wave & 3;  // 2 x 4
  int m0 = tm * 256, n0 = tn * 256;
  int ktiles = K / 64;

  // stage one 128-row half (2 glds/thread): op 0 = A rows, 1 = B rows
  auto stage_half = [&](int kt, int op, int h) {
    int base0 = op ? n0 : m0;
    const T* src = op ? B : A;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int lrow = j * 64 + wave * 8 + (lane >> 3);
      int grow = base0 + h * 128 + lrow;
      int gmax = op ? N - 1 : M - 1;
      if (grow > gmax) grow = gmax;
      const char* g = reinterpret_cast<const char*>(src) +
                      ((long)grow * K + kt * 64) * 2 +
                      (((lane & 7) ^ (lrow & 7)) * 16);
      glds16(g, (char*)lds[kt & 1][op * 2 + h] + (j * 64 + wave * 8) * 128);
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // prologue: tile 0 fully + tile 1's B halves (A(1) stages at t0 q0)
  stage_half(0, 0, 0);
  stage_half(0, 0, 1);
  stage_half(0, 1, 0);
  stage_half(0, 1, 1);
  if (ktiles > 1) {
    stage_half(1, 1, 0);
    stage_half(1, 1, 1);
  }

  for (int kt = 0; kt < ktiles; ++kt) {
    // boundary: allow the 4 most recent loads (2 staged halves) to stay
    // in flight — everything this tile reads landed earlier
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    const char* la = lds[kt & 1][wm];
    const char* lb = lds[kt & 1][2 + (wn >> 1)];
    int bl = (wn & 1) * 64;
    s16x8 bf[2][4];
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      if (q == 0) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            bf[kk][j] = frag_read128(lb, bl + j * 16 + (lane & 15),
                                     (lane >> 4) + kk * 4);
      }
      s16x8 af[2][2];
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int i = 0; i < 2; ++i)
          af[kk][i] = frag_read128(la, q * 32 + i * 16 + (lane & 15),
                                   (lane >> 4) + kk * 4);
      // staggered staging: q0 -> A halves of kt+1; q1/q2 -> B halves of
      // kt+2 (their slots freed by this tile's q0 register loads)
      if (q == 0 && kt + 1 < ktiles) {
        stage_half(kt + 1, 0, 0);
        stage_half(kt + 1, 0, 1);
      } else if (q == 1 && kt + 2 < ktiles) {
        stage_half(kt + 2, 1, 0);
      } else if (q == 2 && kt + 2 < ktiles) {
        stage_half(kt + 2, 1, 1);
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int i = 0; i < 2; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[q * 2 + i][j] =
                MfmaOp<T>::run(af[kk][i], bf[kk][j], acc[q * 2 + i][j]);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
    long rbase = m0 + wm * 128 + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wn * 64 + j * 16 + (lane & 15);
      if (col >= N) continue;
      float badd = BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = rbase + r;
        if (row < M)
          C[(long)row * N + col] = from_f32<OutT>(acc[i][j][r] + badd);
      }
    }
  }
}


// ---- B-direct 256x256 (PDTA_GEMM_BD): only A is glds-staged; B
// fragments are read straight from global memory (per-lane dwordx4 at
// the natural [row][K] layout — no transpose needed for "nt"). The
// 256-col B tile is a few hundred KB and is re-read by every tm block
// of its column panel, so those reads hit the XCD's L2; in exchange the
// kernel stages HALF the bytes through glds — the measured bound of the
// 2-slot schedule (occupancy probe) — and halves its LDS
// footprint (64 KiB -> 3 blocks/CU by LDS).
template <typename T, typename OutT, bool BIAS>
__global__ __launch_bounds__(512) void gemm_nt_bd_k(
    const T* __restrict__ A, const T* __restrict__ B, OutT* __restrict__ C,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n,
    long batch_sa, long batch_sb, long batch_sc) {
  constexpr int BOFF = 256 * 64 * 2;  // A slot bytes
  __shared__ __attribute__((aligned(16))) char lds[2][BOFF];

  A += (long)blockIdx.y * batch_sa;
  B += (long)blockIdx.y * batch_sb;
  C += (long)blockIdx.y * batch_sc;

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 2, wn = wave & 3;  // 2 x 4
  int m0 = tm * 256, n0 = tn * 256;
  int ktiles = K / 64;

  int row_in_chunk = lane >> 3;
  int col16 = lane & 7;

  auto stage_a = [&](int kt, int slot) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = (wave * 4 + i) * 8 + row_in_chunk;
      int grow = m0 + row;
      if (grow > M - 1) grow = M - 1;
      const char* g = reinterpret_cast<const char*>(A) +
                      ((long)grow * K + kt * 64) * 2 +
                      (col16 ^ (row & 7)) * 16;
      glds16(g, (char*)lds[slot] + (wave * 4 + i) * 8 * 128);
    }
  };

  // B fragment rows for this wave (clamped once)
  long brow[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int r = n0 + wn * 64 + j * 16 + (lane & 15);
    if (r > N - 1) r = N - 1;
    brow[j] = (long)r * K;
  }
  int bk = (lane >> 4) * 8;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  stage_a(0, 0);

  for (int kt = 0; kt < ktiles; ++kt) {
    int slot = kt & 1;
    // B fragments for this K-tile: plain vector loads (L2-resident)
    s16x8 bf[2][4];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bf[kk][j] = *reinterpret_cast<const s16x8*>(
            B + brow[j] + kt * 64 + kk * 32 + bk);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (kt + 1 < ktiles) stage_a(kt + 1, slot ^ 1);
    const char* la = lds[slot];
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      s16x8 af[2][2];
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int i = 0; i < 2; ++i)
          af[kk][i] = frag_read128(
              la, wm * 128 + q * 32 + i * 16 + (lane & 15),
              (lane >> 4) + kk * 4);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int i = 0; i < 2; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[q * 2 + i][j] =
                MfmaOp<T>::run(af[kk][i], bf[kk][j], acc[q * 2 + i][j]);
    }
    __builtin_amdgcn_s_barrier();
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
    long rbase = m0 + wm * 128 + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wn * 64 + j * 16 + (lane & 15);
      if (col >= N) continue;
      float badd = BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = rbase + r;
        if (row < M)
          C[(long)row * N + col] = from_f32<OutT>(acc[i][j][r] + badd);
      }
    }
  }
}

// Occupancy experiment (docs/DESIGN.md §6): acc 64 VGPR/thread and
// 2x24 KiB ring slots -> 3 blocks/CU by LDS, ~4 waves/SIMD by VGPR
// (the 256x256/BK=64 kernel is pinned at 2 waves/SIMD and 52% WAIT).

template <typename T, typename OutT, bool BIAS>
__global__ __launch_bounds__(512) void gemm_nt_v2_k(
    const T* __restrict__ A, const T* __restrict__ B, OutT* __restrict__ C,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n,
    long batch_sa, long batch_sb, long batch_sc) {
  constexpr int BOFF = 256 * 64;  // A slot bytes
  __shared__ __attribute__((aligned(16))) char lds[2][BOFF + 128 * 64];

  A += (long)blockIdx.y * batch_sa;
  B += (long)blockIdx.y * batch_sb;
  C += (long)blockIdx.y * batch_sc;

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 1, wn = wave & 1;  // 4 x 2 wave grid of 64x64
  int m0 = tm * 256, n0 = tn * 128;
  int ktiles = K / 32;

  int row_in = lane >> 2, c4 = lane & 3;

  auto stage_all = [&](int kt, int slot) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {  // A: 16 KiB = 16 glds, 2 per wave
      int row = (wave * 2 + i) * 16 + row_in;
      int grow = m0 + row;
      if (grow > M - 1) grow = M - 1;
      const char* g = reinterpret_cast<const char*>(A) +
                      ((long)grow * K + kt * 32) * 2 +
                      ((c4 ^ (row_in & 3)) * 16);
      glds16(g, (char*)lds[slot] + (wave * 2 + i) * 1024);
    }
    {  // B: 8 KiB = 8 glds, 1 per wave
      int row = wave * 16 + row_in;
      int grow = n0 + row;
      if (grow > N - 1) grow = N - 1;
      const char* g = reinterpret_cast<const char*>(B) +
                      ((long)grow * K + kt * 32) * 2 +
                      ((c4 ^ (row_in & 3)) * 16);
      glds16(g, (char*)lds[slot] + BOFF + wave * 1024);
    }
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  stage_all(0, 0);
  for (int kt = 0; kt < ktiles; ++kt) {
    int slot = kt & 1;
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (kt + 1 < ktiles) stage_all(kt + 1, slot ^ 1);
    const char* la = lds[slot];
    const char* lb = lds[slot] + BOFF;
    s16x8 af[4], bf[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      af[i] = frag_read64(la, wm * 64 + i * 16 + (lane & 15), lane >> 4);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      bf[j] = frag_read64(lb, wn * 64 + j * 16 + (lane & 15), lane >> 4);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = MfmaOp<T>::run(af[i], bf[j], acc[i][j]);
    __builtin_amdgcn_s_barrier();
  }

#pragma unroll
  for (int i = 0; i < 4; ++i) {
    long rbase = m0 + wm * 64 + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wn * 64 + j * 16 + (lane & 15);
      if (col >= N) continue;
      float badd = BIAS ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = rbase + r;
        if (row < M)
          C[(long)row * N + col] = from_f32<OutT>(acc[i][j][r] + badd);
      }
    }
  }
}

// vectorized two-stage variant (N % 8 == 0): 16 B loads, 8 columns per
// thread, per-row-slab partials in plain stores + a finalize kernel (an
// atomicAdd version serialized ~700 RMWs per column at the L2 and ran 2x
// SLOWER than the scalar original)
template <typename T>
__global__ void colsum8_part_k(const T* __restrict__ dy,
                               float* __restrict__ part, long M, int N) {
  int c8 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (c8 >= N) return;
  long r0 = blockIdx.y, rs = gridDim.y;
  float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  // 4 independent loads in flight per thread (the rolled loop kept one
  // load outstanding and ran ~5x off the traffic floor)
  long r = r0;
  for (; r + 3 * rs < M; r += 4 * rs) {
    Vec8<T> v0 = load8(dy + r * N + c8);
    Vec8<T> v1 = load8(dy + (r + rs) * N + c8);
    Vec8<T> v2 = load8(dy + (r + 2 * rs) * N + c8);
    Vec8<T> v3 = load8(dy + (r + 3 * rs) * N + c8);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      acc[j] += to_f32(v0.v[j]) + to_f32(v1.v[j]) + to_f32(v2.v[j]) +
                to_f32(v3.v[j]);
  }
  for (; r < M; r += rs) {
    Vec8<T> v = load8(dy + r * N + c8);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += to_f32(v.v[j]);
  }
  float* o = part + (long)blockIdx.y * N + c8;
  *reinterpret_cast<float4*>(o) = *reinterpret_cast<float4*>(acc);
  *reinterpret_cast<float4*>(o + 4) = *reinterpret_cast<float4*>(acc + 4);
}

__global__ void colsum_final_k(const float* __restrict__ part,
                               float* __restrict__ db, int gy, int N) {
  // 16 slices x 64 cols (1024 threads): N=768 is only 12 column-blocks
  // chip-wide, so depth-parallelism inside the block is what keeps this
  // off the latency floor
  __shared__ float red[1024];
  int c = blockIdx.x * 64 + (int)(threadIdx.x % 64);
  int slice = (int)(threadIdx.x / 64);
  float s = 0.f;
  if (c < N)
    for (int k = slice; k < gy; k += 16) s += part[(long)k * N + c];
  red[threadIdx.x] = s;
  __syncthreads();
  if (slice == 0 && c < N) {
#pragma unroll
    for (int i = 1; i < 16; ++i) s += red[threadIdx.x + i * 64];
    db[c] = s;
  }
}

}  // namespace

extern "C" {

// C[M,N] = A[M,K] B[N,K]^T (+bias); K % 64 == 0. out_f32: C is fp32.
void pdta_gemm_nt(const void* A, const void* B, void* C, const float* bias,
                  int M, int N, int K, int dt, int out_f32, hipStream_t s) {
  // occupancy experiment (docs/DESIGN.md §6), opt-in via PDTA_GEMM_V2=1
  static int use_v2 = -1;
  if (use_v2 < 0) use_v2 = getenv("PDTA_GEMM_V2") != nullptr;
  static int use_8p = -1;
  if (use_8p < 0) use_8p = getenv("PDTA_GEMM_8P") != nullptr;
  static int use_r4 = -1;
  if (use_r4 < 0) use_r4 = getenv("PDTA_GEMM_R4") != nullptr;
  static int use_8p2 = -1;
  if (use_8p2 < 0) use_8p2 = getenv("PDTA_GEMM_8P2") != nullptr;
  static int use_bd = -1;
  if (use_bd < 0) use_bd = getenv("PDTA_GEMM_BD") != nullptr;
  if (use_bd && M >= 1024 && N >= 256 && K % 64 == 0 && !out_f32) {
    int tiles_m = (M + 255) / 256, tiles_n = (N + 255) / 256;
    dim3 grid(tiles_m * tiles_n);
    PDTA_DISPATCH_DTYPE16(dt, T, {
      if (bias)
        gemm_nt_bd_k<T, T, true><<<grid, 512, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, bias, M, N, K, tiles_n, 0, 0,
            0);
      else
        gemm_nt_bd_k<T, T, false><<<grid, 512, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, nullptr, M, N, K, tiles_n, 0,
            0, 0);
    });
    return;
  }
  if (use_8p2 && M >= 1024 && N >= 256 && K % 64 == 0 && K >= 256 &&
      !out_f32) {
    int tiles_m = (M + 255) / 256, tiles_n = (N + 255) / 256;
    dim3 grid(tiles_m * tiles_n);
    PDTA_DISPATCH_DTYPE16(dt, T, {
      if (bias)
        gemm_nt_8p2_k<T, T, true><<<grid, 512, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, bias, M, N, K, tiles_n, 0, 0,
            0);
      else
        gemm_nt_8p2_k<T, T, false><<<grid, 512, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, nullptr, M, N, K, tiles_n, 0,
            0, 0);
    });
    return;
  }
  if (use_r4 && M >= 1024 && N >= 256 && K % 32 == 0 && !out_f32) {
    int tiles_m = (M + 255) / 256, tiles_n = (N + 255) / 256;
    dim3 grid(tiles_m * tiles_n);
    PDTA_DISPATCH_DTYPE16(dt, T, {
      if (bias)
        gemm_nt_r4_k<T, T, true><<<grid, 512, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, bias, M, N, K, tiles_n, 0, 0,
            0);
      else
        gemm_nt_r4_k<T, T, false><<<grid, 512, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, nullptr, M, N, K, tiles_n, 0,
            0, 0);
    });
    return;
  }
  if (use_8p && M >= 1024 && N >= 256 && K % 64 == 0 && !out_f32) {
    int tiles_m = (M + 255) / 256, tiles_n = (N + 255) / 256;
    dim3 grid(tiles_m * tiles_n);
    PDTA_DISPATCH_DTYPE16(dt, T, {
      if (bias)
        gemm_nt_8p_k<T, T, true><<<grid, 512, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, bias, M, N, K, tiles_n, 0, 0,
            0);
      else
        gemm_nt_8p_k<T, T, false><<<grid, 512, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, nullptr, M, N, K, tiles_n, 0,
            0, 0);
    });
    return;
  }
  if (use_v2 && M >= 1024 && N >= 128 && K % 32 == 0 && !out_f32) {
    int tiles_m = (M + 255) / 256, tiles_n = (N + 127) / 128;
    dim3 grid(tiles_m * tiles_n);
    PDTA_DISPATCH_DTYPE16(dt, T, {
      if (bias)
        gemm_nt_v2_k<T, T, true><<<grid, 512, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, bias, M, N, K, tiles_n, 0, 0,
            0);
      else
        gemm_nt_v2_k<T, T, false><<<grid, 512, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, nullptr, M, N, K, tiles_n, 0,
            0, 0);
    });
    return;
  }
  if (M >= 1024 && N >= 256 && K % 64 == 0) {
    int tiles_m = (M + 255) / 256, tiles_n = (N + 255) / 256;
    dim3 grid(tiles_m * tiles_n);
    PDTA_DISPATCH_DTYPE16(dt, T, {
      if (out_f32) {
        if (bias)
          gemm_nt_big_k<T, float, true><<<grid, 512, 0, s>>>(
              (const T*)A, (const T*)B, (float*)C, bias, M, N, K, tiles_n,
              0, 0, 0);
        else
          gemm_nt_big_k<T, float, false><<<grid, 512, 0, s>>>(
              (const T*)A, (const T*)B, (float*)C, nullptr, M, N, K,
              tiles_n, 0, 0, 0);
      } else {
        if (bias)
          gemm_nt_big_k<T, T, true><<<grid, 512, 0, s>>>(
              (const T*)A, (const T*)B, (T*)C, bias, M, N, K, tiles_n, 0, 0,
              0);
        else
          gemm_nt_big_k<T, T, false><<<grid, 512, 0, s>>>(
              (const T*)A, (const T*)B, (T*)C, nullptr, M, N, K, tiles_n, 0,
              0, 0);
      }
    });
    return;
  }
  int tiles_m = (M + BM - 1) / BM, tiles_n = (N + BN - 1) / BN;
  dim3 grid(tiles_m * tiles_n);
  PDTA_DISPATCH_DTYPE16(dt, T, {
    if (out_f32) {
      if (bias)
        gemm_nt_k<T, float, true><<<grid, TPB, 0, s>>>(
            (const T*)A, (const T*)B, (float*)C, bias, M, N, K, tiles_n,
            0, 0, 0);
      else
        gemm_nt_k<T, float, false><<<grid, TPB, 0, s>>>(
            (const T*)A, (const T*)B, (float*)C, nullptr, M, N, K, tiles_n,
            0, 0, 0);
    } else {
      if (bias)
        gemm_nt_k<T, T, true><<<grid, TPB, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, bias, M, N, K, tiles_n, 0, 0,
            0);
      else
        gemm_nt_k<T, T, false><<<grid, TPB, 0, s>>>(
            (const T*)A, (const T*)B, (T*)C, nullptr, M, N, K, tiles_n, 0,
            0, 0);
    }
  });
}

// Batched C[b] = A[b] B[b]^T; per-batch element strides. out_f32 option.
void pdta_gemm_nt_batched(const void* A, const void* B, void* C, int M,
                          int N, int K, int batches, long sa, long sb,
                          long sc, int dt, int out_f32, hipStream_t s) {
  int tiles_m = (M + BM - 1) / BM, tiles_n = (N + BN - 1) / BN;
  dim3 grid(tiles_m * tiles_n, batches);
  PDTA_DISPATCH_DTYPE16(dt, T, {
    if (out_f32)
      gemm_nt_k<T, float, false><<<grid, TPB, 0, s>>>(
          (const T*)A, (const T*)B, (float*)C, nullptr, M, N, K, tiles_n,
          sa, sb, sc);
    else
      gemm_nt_k<T, T, false><<<grid, TPB, 0, s>>>(
          (const T*)A, (const T*)B, (T*)C, nullptr, M, N, K, tiles_n, sa,
          sb, sc);
  });
}

// db (fp32 [N]) = column sums of dy [M,N]. part: fp32 workspace of
// pdta_colsum_ws(N) elements (vectorized path); db must be pre-zeroed
// only for the scalar fallback (N % 8 != 0), which atomics into it.
int pdta_colsum_ws(int N) { return N % 8 == 0 ? 512 * N : 0; }

void pdta_colsum_f32(const void* dy, float* db, float* part, long M, int N,
                     int dt, hipStream_t s) {
  dim3 block(256);
  if (N % 8 == 0) {
    long gy = 524288 / N;  // ~64K active lanes in stage 1
    if (gy < 64) gy = 64;
    if (gy > 512) gy = 512;
    if (gy > M) gy = M;
    dim3 grid((N / 8 + 255) / 256, (unsigned)gy);
    PDTA_DISPATCH_DTYPE16(dt, T, {
      colsum8_part_k<T><<<grid, block, 0, s>>>((const T*)dy, part, M, N);
    });
    colsum_final_k<<<(N + 63) / 64, 1024, 0, s>>>(part, db, (int)gy,
                                                  N);
  } else {
    dim3 grid((N + 255) / 256, (unsigned)min((long)512, M));
    PDTA_DISPATCH_DTYPE16(dt, T, {
      colsum_f32_k<T><<<grid, block, 0, s>>>((const T*)dy, db, M, N);
    });
  }
}

}  // extern "C"
