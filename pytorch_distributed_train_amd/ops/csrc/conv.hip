// Implicit-GEMM convolution for gfx950 (SURVEY.md §2d conv rows; replaces
// MIOpen's conv fwd / bwd-data / bwd-weights with NHWC-native MFMA
// kernels — the north-star mandatory hand-written op).
//
// All three directions are GEMM-shaped with K contiguous on both operands
// ("nt" shape, like gemm.hip) thanks to two weight packs:
//   fwd:      y[m=npq][kout]  = sum_{rsc} im2col(x)[m][rsc] * Wt[kout][rsc]
//             Wt pack = [Kout][R,S,C] (c innermost)   -> both K-contiguous
//   bwd-data: dx[m=nhw][c]    = sum_{rs,ko} dyg[m][ko] * W2[r,s,c][ko]
//             W2 pack = [R,S,C][Kout]                 -> both K-contiguous
//   bwd-wei:  dW[kout][c]@rs  = sum_{npq} dy[row][kout] * xg[row][c]
//             transposed LDS fragment reads, fp32 atomics, split-K over npq
// The im2col/dy gathers happen in the per-lane SOURCE address of
// global_load_lds; out-of-bounds taps load from a zero page (128 B of
// zeros) so no lane predication is needed. C and Kout are multiples of 64
// in every non-stem layer, so a 64-wide k-chunk never crosses an (r,s)
// boundary. The C=3 stem (7x7/s2) gets direct kernels (stem_fwd/stem_wrw);
// its dx is never needed (network input).
#include "common.h"
#include "mfma_tile.h"

namespace {

constexpr int TPB = 256;

// ---------------------------------------------------------------- fwd --
// STATS: also emit per-m-tile partial (sum, sumsq) of the f32 output tile
// into bn_part (bn_part_k's [chunk][2C] layout, chunk = tm) so the BN that
// consumes y never re-reads it for statistics (bn.hip pdta_bn_stats stage 1
// eliminated; at B=256 that pass was ~7% of the ResNet-50 step).
template <typename T, int BNT, bool STATS = false>
__global__ __launch_bounds__(TPB) void igemm_fwd_k(
    const T* __restrict__ x, const T* __restrict__ wt, T* __restrict__ y,
    const T* __restrict__ zero_page, int Nn, int H, int W, int C, int P,
    int Q, int Kout, int R, int S, int stride, int pad, int tiles_n,
    float* __restrict__ bn_part = nullptr) {
  // single LDS object (guide §5 trap (a)): A tile 16 KiB + B tile
  constexpr int BOFF = 128 * 64 * 2;
  constexpr int NW_N = (BNT == 128) ? 2 : 1;  // waves along n
  constexpr int NW_M = 4 / NW_N;              // waves along m
  constexpr int MI = 128 / NW_M / 16;         // m-fragments per wave
  constexpr int NJ = BNT / NW_N / 16;         // n-fragments per wave
  __shared__ __attribute__((aligned(16))) char lds[2][BOFF + BNT * 64 * 2];

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave / NW_N, wn = wave % NW_N;
  long M = (long)Nn * P * Q;
  int m0 = tm * 128, n0 = tn * BNT;
  int RSC = R * S * C;

  // staging geometry for this thread's 4 A-rows (and 4 B-rows)
  int row_in_chunk = lane >> 3;
  int gcol16 = (lane & 7);  // swizzle applied per-row below
  int a_n[4], a_h0[4], a_w0[4], a_swz[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row = wave * 32 + i * 8 + row_in_chunk;
    long m = m0 + row;
    if (m > M - 1) m = M - 1;
    int q = (int)(m % Q);
    int p = (int)((m / Q) % P);
    a_n[i] = (int)(m / ((long)P * Q));
    a_h0[i] = p * stride - pad;
    a_w0[i] = q * stride - pad;
    a_swz[i] = gcol16 ^ (row & 7);
  }

  f32x4 acc[MI][NJ];
#pragma unroll
  for (int i = 0; i < MI; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  int cchunks = C / 64;
  int ktiles = R * S * cchunks;

  auto stage = [&](int kt, int buf) {
    int tap = kt / cchunks, ct = kt % cchunks;
    int r = tap / S, ss = tap % S;
    // A: gathered x rows (or zero page)
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int h = a_h0[i] + r, w = a_w0[i] + ss;
      const char* gptr;
      if ((unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W)
        gptr = reinterpret_cast<const char*>(x) +
               ((((long)a_n[i] * H + h) * W + w) * C + ct * 64) * 2 +
               a_swz[i] * 16;
      else
        gptr = reinterpret_cast<const char*>(zero_page);
      glds16(gptr, (char*)lds[buf] + (wave * 32 + i * 8) * 128);
    }
    // B: Wt rows (kout), k = tap*C + ct*64 contiguous
#pragma unroll
    for (int i = 0; i < BNT / 32; ++i) {  // 2 (BNT=64) or 4 (128)
      int row = wave * (BNT / 4) + i * 8 + row_in_chunk;
      int ko = n0 + row;
      if (ko > Kout - 1) ko = Kout - 1;
      const char* gptr = reinterpret_cast<const char*>(wt) +
                         ((long)ko * RSC + tap * C + ct * 64) * 2 +
                         (gcol16 ^ (row & 7)) * 16;
      glds16(gptr, (char*)lds[buf] + BOFF + (wave * (BNT / 4) + i * 8) * 128);
    }
  };

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int kt = 0; kt < ktiles; ++kt) {
    if (kt + 1 < ktiles) stage(kt + 1, cur ^ 1);
    const char* la = lds[cur];
    const char* lb = lds[cur] + BOFF;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      s16x8 af[MI], bf[NJ];
#pragma unroll
      for (int i = 0; i < MI; ++i)
        af[i] = frag_read128(la, wm * (128 / NW_M) + i * 16 + (lane & 15),
                             (lane >> 4) + kk * 4);
#pragma unroll
      for (int j = 0; j < NJ; ++j)
        bf[j] = frag_read128(lb, wn * (BNT / NW_N) + j * 16 + (lane & 15),
                             (lane >> 4) + kk * 4);
#pragma unroll
      for (int i = 0; i < MI; ++i)
#pragma unroll
        for (int j = 0; j < NJ; ++j)
          acc[i][j] = MfmaOp<T>::run(af[i], bf[j], acc[i][j]);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int i = 0; i < MI; ++i) {
    long rbase = m0 + wm * (128 / NW_M) + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < NJ; ++j) {
      int col = n0 + wn * (BNT / NW_N) + j * 16 + (lane & 15);
      if (col >= Kout) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = rbase + r;
        if (row < M) y[row * Kout + col] = from_f32<T>(acc[i][j][r]);
      }
    }
  }

  if (STATS) {
    constexpr int FR = BNT / NW_N;  // columns covered by one wn group
    float* sred = reinterpret_cast<float*>(lds[0]);  // [4 waves][2*FR]
#pragma unroll
    for (int j = 0; j < NJ; ++j) {
      float sj = 0.f, qj = 0.f;
#pragma unroll
      for (int i = 0; i < MI; ++i) {
        long rbase = m0 + wm * (128 / NW_M) + i * 16 + (lane >> 4) * 4;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          if (rbase + r < M) {
            float v = acc[i][j][r];
            sj += v;
            qj += v * v;
          }
      }
      // sum the 4 row-subgroups (lanes l, l+16, l+32, l+48 share a column)
      sj += __shfl_xor(sj, 16);
      sj += __shfl_xor(sj, 32);
      qj += __shfl_xor(qj, 16);
      qj += __shfl_xor(qj, 32);
      if (lane < 16) {
        sred[wave * 2 * FR + j * 16 + lane] = sj;
        sred[wave * 2 * FR + FR + j * 16 + lane] = qj;
      }
    }
    __syncthreads();
    int cpb = Kout < 256 ? Kout : 256;
    for (int t = threadIdx.x; t < BNT; t += TPB) {
      int wn_c = t / FR, off = t % FR;
      float s = 0.f, q = 0.f;
#pragma unroll
      for (int wmi = 0; wmi < NW_M; ++wmi) {
        int wv = wmi * NW_N + wn_c;
        s += sred[wv * 2 * FR + off];
        q += sred[wv * 2 * FR + FR + off];
      }
      int c = n0 + t;
      if (c < Kout) {
        long base = ((long)tm * Kout + (c / cpb) * (long)cpb) * 2;
        bn_part[base + c % cpb] = s;
        bn_part[base + cpb + c % cpb] = q;
      }
    }
  }
}

// ---- big-channel fwd: 256x256 tile, 8 waves, 2-slot K-ring -------------
// The gemm.hip big-tile structure with the im2col gather in the A-source
// (layers with Kout % 256 == 0 carry most of the conv FLOPs; the 128-wide
// tiles ran at ~190 TF from staging re-reads).
template <typename T, bool STATS = false>
__global__ __launch_bounds__(512) void igemm_fwd_big_k(
    const T* __restrict__ x, const T* __restrict__ wt, T* __restrict__ y,
    const T* __restrict__ zero_page, int Nn, int H, int W, int C, int P,
    int Q, int Kout, int R, int S, int stride, int pad, int tiles_n,
    unsigned mq, int sq, unsigned mpq, int spq,
    float* __restrict__ bn_part = nullptr) {
  constexpr int BOFF = 256 * 64 * 2;
  __shared__ __attribute__((aligned(16))) char lds[2][2 * BOFF];

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 2, wn = wave & 3;
  long M = (long)Nn * P * Q;
  int m0 = tm * 256, n0 = tn * 256;
  int RSC = R * S * C;

  int row_in_chunk = lane >> 3;
  int col16 = lane & 7;
  // per-row bases precomputed once; per-stage addresses are base +
  // uniform offset (the 3-mul 64-bit chain per row per stage was ~1/3
  // of this kernel's 8.2-VALU-per-MFMA issue load — PMC r01)
  int a_h0[4], a_w0[4];
  long a_base[4], b_base[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row = (wave * 4 + i) * 8 + row_in_chunk;
    long m = m0 + row;
    if (m > M - 1) m = M - 1;
    unsigned npq = mdiv((unsigned)m, mpq, spq);
    unsigned rem = (unsigned)m - npq * (unsigned)(P * Q);
    unsigned pp = mdiv(rem, mq, sq);
    unsigned qq = rem - pp * (unsigned)Q;
    a_h0[i] = (int)pp * stride - pad;
    a_w0[i] = (int)qq * stride - pad;
    int swz = col16 ^ (row & 7);
    a_base[i] = ((((long)npq * H + a_h0[i]) * W + a_w0[i]) * C) * 2 +
                swz * 16;
    int ko = n0 + row;
    if (ko > Kout - 1) ko = Kout - 1;
    b_base[i] = (long)ko * RSC * 2 + swz * 16;
  }

  int cchunks = C / 64;
  int ktiles = R * S * cchunks;

  auto stage_all = [&](int kt, int slot) {
    int tap = kt / cchunks, ct = kt % cchunks;
    int r = tap / S, ss = tap % S;
    long off_a = ((long)(r * W + ss) * C + ct * 64) * 2;  // uniform
    long off_b = ((long)tap * C + ct * 64) * 2;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int h = a_h0[i] + r, w = a_w0[i] + ss;
      const char* g;
      if ((unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W)
        g = reinterpret_cast<const char*>(x) + a_base[i] + off_a;
      else
        g = reinterpret_cast<const char*>(zero_page);
      glds16(g, (char*)lds[slot] + (wave * 4 + i) * 8 * 128);
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const char* g = reinterpret_cast<const char*>(wt) + b_base[i] + off_b;
      glds16(g, (char*)lds[slot] + BOFF + (wave * 4 + i) * 8 * 128);
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
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
    s16x8 bf[2][4];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bf[kk][j] = frag_read128(lb, wn * 64 + j * 16 + (lane & 15),
                                 (lane >> 4) + kk * 4);
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
      if (col >= Kout) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = rbase + r;
        if (row < M) y[row * Kout + col] = from_f32<T>(acc[i][j][r]);
      }
    }
  }

  if (STATS) {  // per-m-tile (sum, sumsq) partials, see igemm_fwd_k
    float* sred = reinterpret_cast<float*>(lds[0]);  // [8 waves][2*64]
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float sj = 0.f, qj = 0.f;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        long rbase = m0 + wm * 128 + i * 16 + (lane >> 4) * 4;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          if (rbase + r < M) {
            float v = acc[i][j][r];
            sj += v;
            qj += v * v;
          }
      }
      sj += __shfl_xor(sj, 16);
      sj += __shfl_xor(sj, 32);
      qj += __shfl_xor(qj, 16);
      qj += __shfl_xor(qj, 32);
      if (lane < 16) {
        sred[wave * 128 + j * 16 + lane] = sj;
        sred[wave * 128 + 64 + j * 16 + lane] = qj;
      }
    }
    __syncthreads();
    int cpb = Kout < 256 ? Kout : 256;
    for (int t = threadIdx.x; t < 256; t += 512) {
      int wn_c = t / 64, off = t % 64;
      float s = sred[wn_c * 128 + off] + sred[(4 + wn_c) * 128 + off];
      float q = sred[wn_c * 128 + 64 + off] + sred[(4 + wn_c) * 128 + 64 + off];
      int c = n0 + t;
      if (c < Kout) {
        long base = ((long)tm * Kout + (c / cpb) * (long)cpb) * 2;
        bn_part[base + c % cpb] = s;
        bn_part[base + cpb + c % cpb] = q;
      }
    }
  }
}

// ----------------------------------------------------------- bwd-data --
template <typename T, int STRIDE, int BNT, bool ACCUM>  // STRIDE 1/2/0(runtime)
__global__ __launch_bounds__(TPB) void igemm_bwd_data_k(
    const T* __restrict__ dy, const T* __restrict__ w2, T* __restrict__ dx,
    const T* __restrict__ zero_page, int Nn, int H, int W, int C, int P,
    int Q, int Kout, int R, int S, int stride, int pad, int tiles_n,
    const T* __restrict__ accum) {
  // ACCUM as a compile-time split: the runtime `if (accum)` variant cost
  // the no-accum calls 2.5x (measured: big-kernel avg 77 -> 193 us)
  constexpr int BOFF = 128 * 64 * 2;
  constexpr int NW_N = (BNT == 128) ? 2 : 1;
  constexpr int NW_M = 4 / NW_N;
  constexpr int MI = 128 / NW_M / 16;
  constexpr int NJ = BNT / NW_N / 16;
  __shared__ __attribute__((aligned(16))) char lds[2][BOFF + BNT * 64 * 2];

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave / NW_N, wn = wave % NW_N;
  long M = (long)Nn * H * W;
  int m0 = tm * 128, n0 = tn * BNT;

  int row_in_chunk = lane >> 3;
  int gcol16 = (lane & 7);
  int a_n[4], a_h[4], a_w[4], a_swz[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row = wave * 32 + i * 8 + row_in_chunk;
    long m = m0 + row;
    if (m > M - 1) m = M - 1;
    a_w[i] = (int)(m % W);
    a_h[i] = (int)((m / W) % H);
    a_n[i] = (int)(m / ((long)H * W));
    a_swz[i] = gcol16 ^ (row & 7);
  }

  f32x4 acc[MI][NJ];
#pragma unroll
  for (int i = 0; i < MI; ++i)
#pragma unroll
    for (int j = 0; j < NJ; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  int kchunks = Kout / 64;
  int ktiles = R * S * kchunks;

  auto stage = [&](int kt, int buf) {
    int tap = kt / kchunks, kot = kt % kchunks;
    int r = tap / S, ss = tap % S;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int pn = a_h[i] + pad - r;
      int qn = a_w[i] + pad - ss;
      int p, q;
      bool valid;
      if constexpr (STRIDE == 1) {
        p = pn;
        q = qn;
        valid = pn >= 0 && qn >= 0 && p < P && q < Q;
      } else if constexpr (STRIDE == 2) {
        p = pn >> 1;
        q = qn >> 1;
        valid = pn >= 0 && qn >= 0 && !((pn | qn) & 1) && p < P && q < Q;
      } else {
        p = pn / stride;
        q = qn / stride;
        valid = pn >= 0 && qn >= 0 && pn % stride == 0 &&
                qn % stride == 0 && p < P && q < Q;
      }
      const char* gptr;
      if (valid)
        gptr = reinterpret_cast<const char*>(dy) +
               ((((long)a_n[i] * P + p) * Q + q) * Kout + kot * 64) * 2 +
               a_swz[i] * 16;
      else
        gptr = reinterpret_cast<const char*>(zero_page);
      glds16(gptr, (char*)lds[buf] + (wave * 32 + i * 8) * 128);
    }
#pragma unroll
    for (int i = 0; i < BNT / 32; ++i) {
      int row = wave * (BNT / 4) + i * 8 + row_in_chunk;
      int c = n0 + row;
      if (c > C - 1) c = C - 1;
      const char* gptr = reinterpret_cast<const char*>(w2) +
                         ((((long)tap) * C + c) * Kout + kot * 64) * 2 +
                         (gcol16 ^ (row & 7)) * 16;
      glds16(gptr,
             (char*)lds[buf] + BOFF + (wave * (BNT / 4) + i * 8) * 128);
    }
  };

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int kt = 0; kt < ktiles; ++kt) {
    if (kt + 1 < ktiles) stage(kt + 1, cur ^ 1);
    const char* la = lds[cur];
    const char* lb = lds[cur] + BOFF;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      s16x8 af[MI], bf[NJ];
#pragma unroll
      for (int i = 0; i < MI; ++i)
        af[i] = frag_read128(la, wm * (128 / NW_M) + i * 16 + (lane & 15),
                             (lane >> 4) + kk * 4);
#pragma unroll
      for (int j = 0; j < NJ; ++j)
        bf[j] = frag_read128(lb, wn * (BNT / NW_N) + j * 16 + (lane & 15),
                             (lane >> 4) + kk * 4);
#pragma unroll
      for (int i = 0; i < MI; ++i)
#pragma unroll
        for (int j = 0; j < NJ; ++j)
          acc[i][j] = MfmaOp<T>::run(af[i], bf[j], acc[i][j]);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int i = 0; i < MI; ++i) {
    long rbase = m0 + wm * (128 / NW_M) + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < NJ; ++j) {
      int col = n0 + wn * (BNT / NW_N) + j * 16 + (lane & 15);
      if (col >= C) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = rbase + r;
        if (row < M) {
          // ACCUM: residual-fork gradient folded into the store (kills
          // the autograd CUDAFunctor_add pass at every block boundary)
          float v = acc[i][j][r];
          if constexpr (ACCUM) v += to_f32(accum[row * C + col]);
          dx[row * C + col] = from_f32<T>(v);
        }
      }
    }
  }
}

// ---- big-channel bwd-data: 256x256 tile, 2-slot K-ring -----------------
template <typename T, int STRIDE, bool ACCUM>
__global__ __launch_bounds__(512) void igemm_bwd_data_big_k(
    const T* __restrict__ dy, const T* __restrict__ w2, T* __restrict__ dx,
    const T* __restrict__ zero_page, int Nn, int H, int W, int C, int P,
    int Q, int Kout, int R, int S, int stride, int pad, int tiles_n,
    unsigned mw, int sw, unsigned mhw, int shw,
    const T* __restrict__ accum) {
  constexpr int BOFF = 256 * 64 * 2;
  __shared__ __attribute__((aligned(16))) char lds[2][2 * BOFF];

  int bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 2, wn = wave & 3;
  long M = (long)Nn * H * W;
  int m0 = tm * 256, n0 = tn * 256;

  int row_in_chunk = lane >> 3;
  int col16 = lane & 7;
  int a_n[4], a_h[4], a_w[4], a_swz[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row = (wave * 4 + i) * 8 + row_in_chunk;
    long m = m0 + row;
    if (m > M - 1) m = M - 1;
    unsigned nhw = mdiv((unsigned)m, mhw, shw);
    unsigned rem = (unsigned)m - nhw * (unsigned)(H * W);
    unsigned hh = mdiv(rem, mw, sw);
    a_w[i] = (int)(rem - hh * (unsigned)W);
    a_h[i] = (int)hh;
    a_n[i] = (int)nhw;
    a_swz[i] = col16 ^ (row & 7);
  }

  int kchunks = Kout / 64;
  int ktiles = R * S * kchunks;

  auto stage_all = [&](int kt, int slot) {
    int tap = kt / kchunks, kot = kt % kchunks;
    int r = tap / S, ss = tap % S;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int pn = a_h[i] + pad - r;
      int qn = a_w[i] + pad - ss;
      int p, q;
      bool valid;
      if constexpr (STRIDE == 1) {
        p = pn;
        q = qn;
        valid = pn >= 0 && qn >= 0 && p < P && q < Q;
      } else {
        p = pn >> 1;
        q = qn >> 1;
        valid = pn >= 0 && qn >= 0 && !((pn | qn) & 1) && p < P && q < Q;
      }
      const char* g;
      if (valid)
        g = reinterpret_cast<const char*>(dy) +
            ((((long)a_n[i] * P + p) * Q + q) * Kout + kot * 64) * 2 +
            a_swz[i] * 16;
      else
        g = reinterpret_cast<const char*>(zero_page);
      glds16(g, (char*)lds[slot] + (wave * 4 + i) * 8 * 128);
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = (wave * 4 + i) * 8 + row_in_chunk;
      int c = n0 + row;
      if (c > C - 1) c = C - 1;
      const char* g = reinterpret_cast<const char*>(w2) +
                      ((((long)tap) * C + c) * Kout + kot * 64) * 2 +
                      (col16 ^ (row & 7)) * 16;
      glds16(g, (char*)lds[slot] + BOFF + (wave * 4 + i) * 8 * 128);
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
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
    s16x8 bf[2][4];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bf[kk][j] = frag_read128(lb, wn * 64 + j * 16 + (lane & 15),
                                 (lane >> 4) + kk * 4);
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
      if (col >= C) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long row = rbase + r;
        if (row < M) {
          float v = acc[i][j][r];
          if constexpr (ACCUM) v += to_f32(accum[row * C + col]);
          dx[row * C + col] = from_f32<T>(v);
        }
      }
    }
  }
}

// ---------------------------------------------------------- bwd-weight --
// dW[kout][c] at one (r,s): 64x64 tile per block, split-K over npq rows
// (64-row LDS tiles, transposed fragment reads), fp32 atomicAdd into the
// torch-layout dW [Kout,C,R,S] (no repack needed). Rows beyond the image
// or invalid taps stage from the zero page (adds 0).
template <typename T, bool GATHER>  // GATHER=false: 1x1/s1/p0 (identity)
__global__ __launch_bounds__(TPB) void igemm_wrw_k(
    const T* __restrict__ x, const T* __restrict__ dy,
    float* __restrict__ part, const T* __restrict__ zero_page, int Nn,
    int H, int W, int C, int P, int Q, int Kout, int R, int S, int stride,
    int pad, long rows_per_chunk, int ko_tiles, int c_tiles) {
  // 3-deep staging ring with counted vmcnt + raw barrier (guide T3/T4:
  // wrw's compute phase (8 MFMA) is much shorter than the gather's HBM
  // latency, so two tiles stay in flight across each barrier; hipcc's
  // __syncthreads would emit vmcnt(0) and drain them — raw s_barrier only)
  __shared__ __attribute__((aligned(16))) char lds[3][2][64 * 64 * 2];

  int bid = blockIdx.x;
  int tap = bid / (ko_tiles * c_tiles);
  int rem = bid % (ko_tiles * c_tiles);
  int ko_t = rem / c_tiles, c_t = rem % c_tiles;
  int r = tap / S, ss = tap % S;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 1, wn = wave & 1;  // 2x2 waves over 64(ko) x 64(c)

  long NPQ = (long)Nn * P * Q;
  long r0 = (long)blockIdx.y * rows_per_chunk;
  long r1 = min(NPQ, r0 + rows_per_chunk);

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  int row_in_chunk = lane >> 3;
  int col16 = lane & 7;

  // stage 64 rows of dy [row][64 kout] and gathered x [row][64 c]; the
  // source address carries the chunk^=(row&7) swizzle matching the
  // tr_b16 fragment reads (rule 21: glds dest is lane-linear).
  // incremental row->(n,p,q) decode AND incremental byte addresses: the
  // per-tile 64-bit (((n*H+h)*W+w)*C) chains were ~1/3 of this kernel's
  // 18-VALU-per-MFMA issue load (PMC) — all replaced by
  // precomputed per-advance address deltas with bounded carries
  int g_q[2], g_p[2];
  long g_adr[2];  // byte offset of x[n][h][w][c_t*64] (h,w may be OOB)
  int g_h[2], g_w[2];
  const long dq_b = (long)stride * C * 2;          // q += 1
  const long dp_b = (long)stride * W * C * 2;      // p += 1
  const long dn_b = (long)H * W * C * 2;           // n += 1
  if (GATHER) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      long row = r0 + wave * 16 + i * 8 + row_in_chunk;
      g_q[i] = (int)(row % Q);
      g_p[i] = (int)((row / Q) % P);
      int n = (int)(row / ((long)P * Q));
      g_h[i] = g_p[i] * stride - pad + r;
      g_w[i] = g_q[i] * stride - pad + ss;
      g_adr[i] = (long)n * dn_b + (long)g_h[i] * W * C * 2 +
                 (long)g_w[i] * C * 2;
    }
  }
  int adv_dq = 64 % Q, adv_dp = 64 / Q;  // bounded carries per advance
  auto advance = [&]() {
    if (!GATHER) return;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      g_q[i] += adv_dq;
      int cy = g_q[i] >= Q;
      g_q[i] -= cy ? Q : 0;
      int dp = adv_dp + cy;
      g_p[i] += dp;
      g_w[i] += (adv_dq - (cy ? Q : 0)) * stride;
      g_h[i] += dp * stride;
      g_adr[i] += (long)(adv_dq - (cy ? Q : 0)) * dq_b + (long)dp * dp_b;
      while (g_p[i] >= P) {
        g_p[i] -= P;
        g_h[i] -= P * stride;
        g_adr[i] += dn_b - (long)P * dp_b;
      }
    }
  };
  // dy addresses advance by a constant per tile; the swizzled column
  // offset is per-i constant — both hoisted out of the loop
  const char* gdy_p[2];
  const char* gx_sw[2];
  long dy_row[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int ldsrow = wave * 16 + i * 8 + row_in_chunk;
    int scol16 = col16 ^ (ldsrow & 7);
    dy_row[i] = r0 + ldsrow;
    gdy_p[i] = reinterpret_cast<const char*>(dy) +
               (dy_row[i] * Kout + ko_t * 64) * 2 + scol16 * 16;
    gx_sw[i] = reinterpret_cast<const char*>(x) +
               ((long)c_t * 64) * 2 + scol16 * 16;
  }
  const long dy_step = (long)64 * Kout * 2;
  auto stage = [&](long rowbase, int buf) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int ldsrow = wave * 16 + i * 8 + row_in_chunk;
      const char* gdy;
      const char* gx;
      if (dy_row[i] < r1) {
        gdy = gdy_p[i];
        if (GATHER) {
          if ((unsigned)g_h[i] < (unsigned)H &&
              (unsigned)g_w[i] < (unsigned)W)
            gx = gx_sw[i] + g_adr[i];
          else
            gx = reinterpret_cast<const char*>(zero_page);
        } else {
          gx = gx_sw[i] + dy_row[i] * (long)C * 2;
        }
      } else {
        gdy = reinterpret_cast<const char*>(zero_page);
        gx = reinterpret_cast<const char*>(zero_page);
      }
      glds16(gdy, (char*)lds[buf][0] + (wave * 16 + i * 8) * 128);
      glds16(gx, (char*)lds[buf][1] + (wave * 16 + i * 8) * 128);
      gdy_p[i] += dy_step;
      dy_row[i] += 64;
    }
  };

  // prologue: two tiles in flight (each stage = 4 glds per wave)
  stage(r0, 0);
  if (r0 + 64 < r1) {
    advance();
    stage(r0 + 64, 1);
  }
  unsigned ldy_base[3] = {lds_addr_of(lds[0][0]), lds_addr_of(lds[1][0]),
                          lds_addr_of(lds[2][0])};
  unsigned lx_base[3] = {lds_addr_of(lds[0][1]), lds_addr_of(lds[1][1]),
                         lds_addr_of(lds[2][1])};
  // loop-invariant tr-read offsets (8 fragment pairs) precomputed once
  unsigned a_off0[2][2], a_off1[2][2], b_off0[2][2], b_off1[2][2];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk)
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      tr_off(kk * 32, wm * 32 + i * 16, lane, &a_off0[kk][i],
             &a_off1[kk][i]);
      tr_off(kk * 32, wn * 32 + i * 16, lane, &b_off0[kk][i],
             &b_off1[kk][i]);
    }
  int cur = 0;
  for (long rb = r0; rb < r1; rb += 64) {
    // wait for THIS tile's 4 glds (<=4 others stay in flight), then a raw
    // barrier so every wave's rows of tile `cur` have landed
    if (rb + 64 < r1)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    // issue ALL 16 transpose reads (both K=32 steps), wait once, then run
    // the 8 MFMAs — the reads' latencies overlap instead of chaining
    unsigned long long alo[2][2], ahi[2][2], blo[2][2], bhi[2][2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int i = 0; i < 2; ++i)
        tr_issue_at(ldy_base[cur] + a_off0[kk][i],
                    ldy_base[cur] + a_off1[kk][i], &alo[kk][i],
                    &ahi[kk][i]);
#pragma unroll
      for (int i = 0; i < 2; ++i)
        tr_issue_at(lx_base[cur] + b_off0[kk][i],
                    lx_base[cur] + b_off1[kk][i], &blo[kk][i],
                    &bhi[kk][i]);
    }
    tr_read_wait();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      s16x8 af[2], bf[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        af[i] = tr_pair_to_frag(alo[kk][i], ahi[kk][i]);
        bf[i] = tr_pair_to_frag(blo[kk][i], bhi[kk][i]);
      }
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = MfmaOp<T>::run(af[i], bf[j], acc[i][j]);
    }
    // all waves done READING tile cur before its buffer is re-staged two
    // iterations from now (the next loop's barrier alone would only cover
    // one iteration) — raw barrier, no vmcnt drain
    __builtin_amdgcn_s_barrier();
    if (rb + 128 < r1) {
      advance();
      stage(rb + 128, (cur + 2) % 3);
    }
    cur = (cur + 1) % 3;
  }

  // write this chunk's 64x64 fp32 partial tile (coalesced; a final
  // reduce kernel sums chunks into dW — global atomicAdd here put up to
  // ~450 RMWs on every dW element and dominated the kernel)
  float* out = part + ((long)blockIdx.y * gridDim.x + blockIdx.x) * 4096;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int c_local = wn * 32 + j * 16 + (lane & 15);
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        int ko_local = wm * 32 + i * 16 + (lane >> 4) * 4 + rg;
        out[ko_local * 64 + c_local] = acc[i][j][rg];
      }
    }
  }
}

// sum the per-chunk partials into dW's torch layout [Kout,C,R,S]
// first-level slab reduce for long chunk lists: [chunks][tiles*4096] ->
// [16][tiles*4096] (small-tile layers run splitk up to 256; a single
// reduce pass there had only tiles*16 waves, each chunk-serial)
__global__ void wrw_mid_k(const float* __restrict__ part,
                          float* __restrict__ out, int chunks, long total4) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride_ = (long)gridDim.x * blockDim.x;
  int slab = blockIdx.y;  // 16 slabs
  const float4* p4 = reinterpret_cast<const float4*>(part);
  float4* o4 = reinterpret_cast<float4*>(out);
  for (; idx < total4; idx += stride_) {
    float4 sv = {0.f, 0.f, 0.f, 0.f};
    for (int k = slab; k < chunks; k += 16) {
      float4 v = p4[(long)k * total4 + idx];
      sv.x += v.x;
      sv.y += v.y;
      sv.z += v.z;
      sv.w += v.w;
    }
    o4[(long)slab * total4 + idx] = sv;
  }
}

__global__ void wrw_reduce_k(const float* __restrict__ part,
                             float* __restrict__ dw, int chunks, int tiles,
                             int ko_tiles, int c_tiles, int C, int Kout,
                             int R, int S) {
  // 4 consecutive c-elements per thread: 16-B slab loads (a scalar-f32
  // version was ~3.7x off the traffic floor in rocprof)
  long total4 = (long)tiles * 1024;
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride_ = (long)gridDim.x * blockDim.x;
  for (; idx < total4; idx += stride_) {
    int tile = (int)(idx >> 10);
    int local = (int)(idx & 1023) * 4;
    float4 sv = {0.f, 0.f, 0.f, 0.f};
    const float* p = part + (long)tile * 4096 + local;
    long tstride = (long)tiles * 4096;
    for (int k = 0; k < chunks; ++k) {
      float4 v = *reinterpret_cast<const float4*>(p + (long)k * tstride);
      sv.x += v.x;
      sv.y += v.y;
      sv.z += v.z;
      sv.w += v.w;
    }
    int tap = tile / (ko_tiles * c_tiles);
    int rem = tile % (ko_tiles * c_tiles);
    int ko_t = rem / c_tiles, c_t = rem % c_tiles;
    int r = tap / S, ss = tap % S;
    int ko = ko_t * 64 + (local >> 6);
    int c = c_t * 64 + (local & 63);
    const float* sp = reinterpret_cast<const float*>(&sv);
    if (ko < Kout) {
      // each (ko,c,r,s) is produced by exactly one (tile,lane): plain
      // store, so dw needs no zero-fill before the reduce
#pragma unroll
      for (int j = 0; j < 4; ++j)
        if (c + j < C)
          dw[(((long)ko * C + c + j) * R + r) * S + ss] = sp[j];
    }
  }
}


// ---- medium bwd-weight: 128x128 (ko x c) tile, 4 waves, 3-slot ring.
// For Kout%128==0 && C%128==0 shapes below the big kernel's gate: the
// 64x64 tile re-stages every dy row c_tiles(=C/64) times and every x row
// ko_tiles times; this tile HALVES both (measured: the wrw family was
// 21%% of the ResNet-50 step, gather-staging bound). Partials land as
// 64x64 granules in wrw_reduce_k's layout (4 granules per block).
template <typename T, bool GATHER>
__global__ __launch_bounds__(TPB) void igemm_wrw_mid_k(
    const T* __restrict__ x, const T* __restrict__ dy,
    float* __restrict__ part, const T* __restrict__ zero_page, int Nn,
    int H, int W, int C, int P, int Q, int Kout, int R, int S, int stride,
    int pad, long rows_per_chunk, int ko_tiles, int c_tiles,
    long tot_granules) {
  // slots: dy [64 rows][128 ko] + x [64 rows][128 c], 256-B rows
  __shared__ __attribute__((aligned(16))) char lds[3][2][64 * 128 * 2];

  int ko_t2 = ko_tiles / 2, c_t2 = c_tiles / 2;
  int bid = blockIdx.x;
  int tap = bid / (ko_t2 * c_t2);
  int rem = bid % (ko_t2 * c_t2);
  int ko_t = rem / c_t2, c_t = rem % c_t2;  // in 128-units
  int r = tap / S, ss = tap % S;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 1, wn = wave & 1;  // 2x2 waves over 128ko x 128c

  long NPQ = (long)Nn * P * Q;
  long r0 = (long)blockIdx.y * rows_per_chunk;
  long r1 = min(NPQ, r0 + rows_per_chunk);

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  int row4 = lane >> 4;       // 4 rows per glds (256-B rows, 16 chunks)
  int chunk16 = lane & 15;

  long g_adr[4];
  int g_h[4], g_w[4];
  const long dq_b = (long)stride * C * 2;
  const long dp_b = (long)stride * W * C * 2;
  const long dn_b = (long)H * W * C * 2;
  int gq4[4], gp4[4];
  if (GATHER) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      long row = r0 + wave * 16 + i * 4 + row4;
      gq4[i] = (int)(row % Q);
      gp4[i] = (int)((row / Q) % P);
      int n = (int)(row / ((long)P * Q));
      g_h[i] = gp4[i] * stride - pad + r;
      g_w[i] = gq4[i] * stride - pad + ss;
      g_adr[i] = (long)n * dn_b + (long)g_h[i] * W * C * 2 +
                 (long)g_w[i] * C * 2;
    }
  }
  int adv_dq = 64 % Q, adv_dp = 64 / Q;
  auto advance = [&]() {
    if (!GATHER) return;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      gq4[i] += adv_dq;
      int cy = gq4[i] >= Q;
      gq4[i] -= cy ? Q : 0;
      int dp = adv_dp + cy;
      gp4[i] += dp;
      g_w[i] += (adv_dq - (cy ? Q : 0)) * stride;
      g_h[i] += dp * stride;
      g_adr[i] += (long)(adv_dq - (cy ? Q : 0)) * dq_b + (long)dp * dp_b;
      while (gp4[i] >= P) {
        gp4[i] -= P;
        g_h[i] -= P * stride;
        g_adr[i] += dn_b - (long)P * dp_b;
      }
    }
  };
  const char* gdy_p[4];
  const char* gx_sw[4];
  long dy_row[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int ldsrow = wave * 16 + i * 4 + row4;
    int scol = chunk16 ^ (ldsrow & 15);
    dy_row[i] = r0 + ldsrow;
    gdy_p[i] = reinterpret_cast<const char*>(dy) +
               (dy_row[i] * Kout + (long)ko_t * 128) * 2 + scol * 16;
    gx_sw[i] = reinterpret_cast<const char*>(x) +
               ((long)c_t * 128) * 2 + scol * 16;
  }
  const long dy_step = (long)64 * Kout * 2;
  auto stage = [&](int buf) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const char* gdy;
      const char* gx;
      if (dy_row[i] < r1) {
        gdy = gdy_p[i];
        if (GATHER) {
          if ((unsigned)g_h[i] < (unsigned)H &&
              (unsigned)g_w[i] < (unsigned)W)
            gx = gx_sw[i] + g_adr[i];
          else
            gx = reinterpret_cast<const char*>(zero_page);
        } else {
          gx = gx_sw[i] + dy_row[i] * (long)C * 2;
        }
      } else {
        gdy = reinterpret_cast<const char*>(zero_page);
        gx = reinterpret_cast<const char*>(zero_page);
      }
      // glds dest is the 4-row group base (lane-linear fill)
      glds16(gdy, (char*)lds[buf][0] + (wave * 16 + i * 4) * 256);
      glds16(gx, (char*)lds[buf][1] + (wave * 16 + i * 4) * 256);
      gdy_p[i] += dy_step;
      dy_row[i] += 64;
    }
  };

  stage(0);
  if (r0 + 64 < r1) {
    advance();
    stage(1);
  }
  unsigned ldy_base[3] = {lds_addr_of(lds[0][0]), lds_addr_of(lds[1][0]),
                          lds_addr_of(lds[2][0])};
  unsigned lx_base[3] = {lds_addr_of(lds[0][1]), lds_addr_of(lds[1][1]),
                         lds_addr_of(lds[2][1])};
  unsigned a_off0[2][2], a_off1[2][2], b_off0[2][2], b_off1[2][2];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk)
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      tr_off_s<256>(kk * 32, wm * 64 + i * 16 + 0, lane, &a_off0[kk][i],
                    &a_off1[kk][i]);
      tr_off_s<256>(kk * 32, wn * 64 + i * 16 + 0, lane, &b_off0[kk][i],
                    &b_off1[kk][i]);
    }
  // each wave computes 64ko x 64c = (2 a-frag groups of 32) x ... the
  // 4x4 acc spans 64x64 via two 16-row a-frags and two 16-col b-frags
  // per kk, with a second (i+2, j+2) pass offset by 32 rows/cols
  unsigned a_off0b[2][2], a_off1b[2][2], b_off0b[2][2], b_off1b[2][2];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk)
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      tr_off_s<256>(kk * 32, wm * 64 + 32 + i * 16, lane, &a_off0b[kk][i],
                    &a_off1b[kk][i]);
      tr_off_s<256>(kk * 32, wn * 64 + 32 + i * 16, lane, &b_off0b[kk][i],
                    &b_off1b[kk][i]);
    }
  int cur = 0;
  for (long rb = r0; rb < r1; rb += 64) {
    if (rb + 64 < r1)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    unsigned long long alo[2][4], ahi[2][4], blo[2][4], bhi[2][4];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        tr_issue_at(ldy_base[cur] + a_off0[kk][i],
                    ldy_base[cur] + a_off1[kk][i], &alo[kk][i],
                    &ahi[kk][i]);
        tr_issue_at(ldy_base[cur] + a_off0b[kk][i],
                    ldy_base[cur] + a_off1b[kk][i], &alo[kk][i + 2],
                    &ahi[kk][i + 2]);
        tr_issue_at(lx_base[cur] + b_off0[kk][i],
                    lx_base[cur] + b_off1[kk][i], &blo[kk][i],
                    &bhi[kk][i]);
        tr_issue_at(lx_base[cur] + b_off0b[kk][i],
                    lx_base[cur] + b_off1b[kk][i], &blo[kk][i + 2],
                    &bhi[kk][i + 2]);
      }
    }
    tr_read_wait();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      s16x8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        af[i] = tr_pair_to_frag(alo[kk][i], ahi[kk][i]);
        bf[i] = tr_pair_to_frag(blo[kk][i], bhi[kk][i]);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = MfmaOp<T>::run(af[i], bf[j], acc[i][j]);
    }
    __builtin_amdgcn_s_barrier();
    if (rb + 128 < r1) {
      advance();
      stage((cur + 2) % 3);
    }
    cur = (cur + 1) % 3;
  }

  // Wave (wm,wn) owns rows wm*64..+63, cols wn*64..+63 of the 128x128
  // block = exactly ONE 64x64 granule: (ko64, c64) = (ko_t*2+wm, c_t*2+wn)
  {
    long g64 = (long)tap * ko_tiles * c_tiles +
               (long)(ko_t * 2 + wm) * c_tiles + (c_t * 2 + wn);
    float* out = part + ((long)blockIdx.y * tot_granules + g64) * 4096;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int ihalf = (i >= 2);     // af index: 0,1 -> +0; 2,3 -> +32
      int ko_local = ihalf * 32 + (i & 1) * 16 + (lane >> 4) * 4;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int jhalf = (j >= 2);
        int c_local = jhalf * 32 + (j & 1) * 16 + (lane & 15);
#pragma unroll
        for (int rg = 0; rg < 4; ++rg)
          out[(ko_local + rg) * 64 + c_local] = acc[i][j][rg];
      }
    }
  }
}

// ---- big bwd-weight: 256x256 (ko x c) tile, 8 waves, 2-slot ring -------
// For Kout % 256 == 0 && C % 256 == 0 (including the linear-dW TN GEMMs
// routed through this path): the 64x64-tile kernel re-stages every row
// once per output tile — 16x the staged bytes of this one. Tiles are
// [64 rows][256 features] (512-B rows), both fragments hardware-transposed
// (ds_read_b64_tr_b16). Partials land as 64x64 granules in the SAME
// layout as the small kernel, so wrw_reduce_k is shared.
template <typename T, bool GATHER>
__global__ __launch_bounds__(512) void igemm_wrw_big_k(
    const T* __restrict__ x, const T* __restrict__ dy,
    float* __restrict__ part, const T* __restrict__ zero_page, int Nn,
    int H, int W, int C, int P, int Q, int Kout, int R, int S, int stride,
    int pad, long rows_per_chunk, int ko_tiles, int c_tiles,
    unsigned mq, int sq, unsigned mpq, int spq) {
  constexpr int BOFF = 64 * 256 * 2;  // one operand image per slot
  __shared__ __attribute__((aligned(16))) char lds[2][2 * BOFF];

  int nbk = ko_tiles / 4, nbc = c_tiles / 4;  // 256-wide big tiles
  int tap = blockIdx.x / (nbk * nbc);
  int rem = blockIdx.x % (nbk * nbc);
  int bko = rem / nbc, bc = rem % nbc;
  int r = tap / S, ss = tap % S;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 2, wn = wave & 3;  // 2 x 4 over (256 ko x 256 c)

  long NPQ = (long)Nn * P * Q;
  long r0 = (long)blockIdx.y * rows_per_chunk;
  long r1 = min(NPQ, r0 + rows_per_chunk);
  if (r0 >= r1) return;

  auto stage_all = [&](long rowbase, int slot) {
    // each glds: 64 lanes x 16B = 2 rows of 512 B; 4 glds/wave x 8 waves
    // = the full 64-row image per operand
    int lrow_in = lane >> 5;   // 0..1 within the glds' 2 rows
    int c32 = lane & 31;       // 16-B chunk within the 512-B row
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int lrow = (wave * 4 + i) * 2 + lrow_in;
      long row = rowbase + lrow;
      int swz = (c32 ^ (lrow & 31)) * 16;  // full-range XOR (see
      // frag_read_tr_issue_s)
      const char* g;
      if (row < r1)
        g = reinterpret_cast<const char*>(dy) +
            (row * Kout + bko * 256) * 2 + swz;
      else
        g = reinterpret_cast<const char*>(zero_page);
      glds16(g, (char*)lds[slot] + (wave * 4 + i) * 1024);
    }
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int lrow = (wave * 4 + i) * 2 + lrow_in;
      long row = rowbase + lrow;
      int swz = (c32 ^ (lrow & 31)) * 16;  // full-range XOR (see
      // frag_read_tr_issue_s)
      const char* g = reinterpret_cast<const char*>(zero_page);
      if (row < r1) {
        if (GATHER) {
          unsigned npq = mdiv((unsigned)row, mpq, spq);
          unsigned rm = (unsigned)row - npq * (unsigned)(P * Q);
          unsigned pp = mdiv(rm, mq, sq);
          unsigned qq = rm - pp * (unsigned)Q;
          int h = (int)pp * stride - pad + r;
          int w = (int)qq * stride - pad + ss;
          if ((unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W)
            g = reinterpret_cast<const char*>(x) +
                ((((long)npq * H + h) * W + w) * C + bc * 256) * 2 + swz;
        } else {
          g = reinterpret_cast<const char*>(x) +
              (row * C + bc * 256) * 2 + swz;
        }
      }
      glds16(g, (char*)lds[slot] + BOFF + (wave * 4 + i) * 1024);
    }
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  stage_all(r0, 0);
  unsigned dy_base[2] = {lds_addr_of(lds[0]), lds_addr_of(lds[1])};
  for (long rb = r0; rb < r1; rb += 64) {
    int slot = (int)((rb - r0) >> 6) & 1;
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (rb + 64 < r1) stage_all(rb + 64, slot ^ 1);
    unsigned la = dy_base[slot];
    unsigned lb = dy_base[slot] + BOFF;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      unsigned long long alo[8], ahi[8], blo[4], bhi[4];
#pragma unroll
      for (int i = 0; i < 8; ++i)
        frag_read_tr_issue_s<512>(la, kk * 32, wm * 128 + i * 16, lane,
                                  &alo[i], &ahi[i]);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        frag_read_tr_issue_s<512>(lb, kk * 32, wn * 64 + j * 16, lane,
                                  &blo[j], &bhi[j]);
      tr_read_wait();
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        s16x8 af = tr_pair_to_frag(alo[i], ahi[i]);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = MfmaOp<T>::run(af, tr_pair_to_frag(blo[j], bhi[j]),
                                     acc[i][j]);
      }
    }
    __builtin_amdgcn_s_barrier();
  }

  // write 4x4 granules of 64x64 fp32 in the shared wrw partial layout
  int tiles_total = ko_tiles * c_tiles * R * S;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int ko_local = wm * 128 + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int c_local = wn * 64 + j * 16 + (lane & 15);
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        int kol = ko_local + rg;
        int t64 = tap * ko_tiles * c_tiles +
                  (bko * 4 + (kol >> 6)) * c_tiles + (bc * 4 + (c_local >> 6));
        float* out =
            part + ((long)blockIdx.y * tiles_total + t64) * 4096;
        out[(kol & 63) * 64 + (c_local & 63)] = acc[i][j][rg];
      }
    }
  }
}

// ------------------------------------------------------------- stem ----
// The C=3 stem (7x7/s2) goes through explicit im2col into the SAME MFMA
// igemm path as every other conv: xcol[npq][rsc_pad] with rsc = R*S*Cin
// zero-padded to a multiple of 64 (zero columns match zero-padded weight
// packs, so results are exact). The direct stem kernels this replaces
// were VALU-issue- and latency-bound (stem wrw alone was 29-40% of the
// ResNet-50 step — profiles/r50_b256_kernel_stats_r01.md).
template <typename T>
__global__ void stem_im2col_k(const T* __restrict__ x, T* __restrict__ out,
                              int Nn, int H, int W, int Cin, int P, int Q,
                              int R, int S, int stride, int pad, int rsc_pad,
                              unsigned mq, int sq, unsigned mpq, int spq,
                              unsigned mc, int sc_, unsigned msc, int ssc) {
  // item = (row, 8-element group): consecutive threads write consecutive
  // 16-B chunks of one row (fully coalesced); row decode via magic
  // division (runtime-divisor div was ~40% of this kernel's issue slots)
  int g8 = rsc_pad / 8;
  long total = (long)Nn * P * Q * g8;
  long item = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long gstride = (long)gridDim.x * blockDim.x;
  int rsc = R * S * Cin;
  for (; item < total; item += gstride) {
    int grp = (int)(item % g8);  // g8 is 24 for the stem: strength-reduced?
    unsigned row = (unsigned)(item / g8);
    unsigned npq = mdiv(row, mpq, spq);      // row / (P*Q)
    unsigned rem = row - npq * (unsigned)(P * Q);
    unsigned p = mdiv(rem, mq, sq);          // rem / Q
    unsigned q = rem - p * (unsigned)Q;
    int n = (int)npq;
    int h0 = (int)p * stride - pad, w0 = (int)q * stride - pad;
    Vec8<T> v;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int e = grp * 8 + j;
      float val = 0.f;
      if (e < rsc) {
        int t = (int)mdiv((unsigned)e, mc, sc_);      // e / Cin
        int c = e - t * Cin;
        int rr = (int)mdiv((unsigned)t, msc, ssc);    // t / S
        int sc = t - rr * S;
        int h = h0 + rr, w = w0 + sc;
        if ((unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W)
          val = to_f32(x[(((long)n * H + h) * W + w) * Cin + c]);
      }
      v.v[j] = from_f32<T>(val);
    }
    store8(out + (long)row * rsc_pad + grp * 8, v);
  }
}

// v2: for [r][s][c]-ordered patches over NHWC input, the S*Cin elements
// of one tap-row are CONTIGUOUS in x — copy them as a run instead of
// decoding every element (the v1 per-element decode ran 7.5x off the
// write-traffic floor). Task R zeroes the rsc..rsc_pad tail (the padded
// GEMM K columns must be 0, not garbage: 0*NaN poisons the output).
template <typename T>
__global__ void stem_im2col2_k(const T* __restrict__ x, T* __restrict__ out,
                               int Nn, int H, int W, int Cin, int P, int Q,
                               int R, int S, int stride, int pad,
                               int rsc_pad, unsigned mq, int sq,
                               unsigned mpq, int spq) {
  int L = S * Cin;
  int rsc = R * S * Cin;
  int tasks = R + 1;
  long total = (long)Nn * P * Q * tasks;
  long item = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long gstride = (long)gridDim.x * blockDim.x;
  for (; item < total; item += gstride) {
    int task = (int)(item % tasks);
    unsigned row = (unsigned)(item / tasks);
    unsigned npq = mdiv(row, mpq, spq);
    unsigned rem = row - npq * (unsigned)(P * Q);
    unsigned p = mdiv(rem, mq, sq);
    unsigned q = rem - p * (unsigned)Q;
    T* orow = out + (long)row * rsc_pad;
    if (task == R) {
      for (int e = rsc; e < rsc_pad; ++e) orow[e] = from_f32<T>(0.f);
      continue;
    }
    int h = (int)p * stride - pad + task;
    int w0 = (int)q * stride - pad;
    T* dst = orow + task * L;
    if ((unsigned)h < (unsigned)H && w0 >= 0 && w0 + S <= W) {
      const T* src = x + (((long)npq * H + h) * W + w0) * Cin;
      for (int e = 0; e < L; ++e) dst[e] = src[e];
    } else {
      for (int s2 = 0; s2 < S; ++s2) {
        int w = w0 + s2;
        bool ok = (unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W;
        const T* src = x + (((long)npq * H + h) * W + w) * Cin;
        for (int c = 0; c < Cin; ++c)
          dst[s2 * Cin + c] = ok ? src[c] : from_f32<T>(0.f);
      }
    }
  }
}

}  // namespace

extern "C" {

// m-tile count of the kernel pdta_conv_fwd will pick = the chunk count of
// the bn_part slab it writes when bn_part != nullptr
long pdta_conv_fwd_chunks(long M, int Kout) {
  if (Kout % 256 == 0 && M >= 4096) return (M + 255) / 256;
  return (M + 127) / 128;
}

void pdta_conv_fwd(const void* x, const void* wt, void* y,
                   const void* zero_page, int Nn, int H, int W, int C, int P,
                   int Q, int Kout, int R, int S, int stride, int pad,
                   int dt, hipStream_t s, float* bn_part) {
  long M = (long)Nn * P * Q;
  if (Kout % 256 == 0 && M >= 4096) {
    int tiles_m = (int)((M + 255) / 256), tiles_n = Kout / 256;
    dim3 grid(tiles_m * tiles_n);
    MagicDiv mq = magic_div(Q), mpq = magic_div(P * Q);
    PDTA_DISPATCH_DTYPE16(dt, T, {
      if (bn_part)
        igemm_fwd_big_k<T, true><<<grid, 512, 0, s>>>(
            (const T*)x, (const T*)wt, (T*)y, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, tiles_n, mq.mul, mq.shift,
            mpq.mul, mpq.shift, bn_part);
      else
        igemm_fwd_big_k<T><<<grid, 512, 0, s>>>(
            (const T*)x, (const T*)wt, (T*)y, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, tiles_n, mq.mul, mq.shift,
            mpq.mul, mpq.shift);
    });
    return;
  }
  int bnt = (Kout % 128 == 0) ? 128 : 64;
  int tiles_m = (int)((M + 127) / 128), tiles_n = (Kout + bnt - 1) / bnt;
  dim3 grid(tiles_m * tiles_n);
  PDTA_DISPATCH_DTYPE16(dt, T, {
    if (bnt == 128) {
      if (bn_part)
        igemm_fwd_k<T, 128, true><<<grid, TPB, 0, s>>>(
            (const T*)x, (const T*)wt, (T*)y, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, tiles_n, bn_part);
      else
        igemm_fwd_k<T, 128><<<grid, TPB, 0, s>>>(
            (const T*)x, (const T*)wt, (T*)y, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, tiles_n);
    } else {
      if (bn_part)
        igemm_fwd_k<T, 64, true><<<grid, TPB, 0, s>>>(
            (const T*)x, (const T*)wt, (T*)y, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, tiles_n, bn_part);
      else
        igemm_fwd_k<T, 64><<<grid, TPB, 0, s>>>(
            (const T*)x, (const T*)wt, (T*)y, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, tiles_n);
    }
  });
}

void pdta_conv_bwd_data(const void* dy, const void* w2, void* dx,
                        const void* zero_page, const void* accum, int Nn,
                        int H, int W, int C, int P, int Q, int Kout, int R,
                        int S, int stride, int pad, int dt,
                        hipStream_t s) {
  long M = (long)Nn * H * W;
  if (C % 256 == 0 && M >= 4096 && (stride == 1 || stride == 2)) {
    int tiles_m = (int)((M + 255) / 256), tiles_n = C / 256;
    dim3 grid(tiles_m * tiles_n);
    MagicDiv mw = magic_div(W), mhw = magic_div(H * W);
#define PDTA_BWD_BIG(ST, ACC)                                             \
      igemm_bwd_data_big_k<T, ST, ACC><<<grid, 512, 0, s>>>(              \
          (const T*)dy, (const T*)w2, (T*)dx, (const T*)zero_page, Nn, H, \
          W, C, P, Q, Kout, R, S, stride, pad, tiles_n, mw.mul, mw.shift, \
          mhw.mul, mhw.shift, (const T*)accum)
    PDTA_DISPATCH_DTYPE16(dt, T, {
      if (stride == 1) {
        if (accum) PDTA_BWD_BIG(1, true); else PDTA_BWD_BIG(1, false);
      } else {
        if (accum) PDTA_BWD_BIG(2, true); else PDTA_BWD_BIG(2, false);
      }
    });
#undef PDTA_BWD_BIG
    return;
  }
  int bnt = (C % 128 == 0) ? 128 : 64;
  int tiles_m = (int)((M + 127) / 128), tiles_n = (C + bnt - 1) / bnt;
  dim3 grid(tiles_m * tiles_n);
#define PDTA_BWD_SMALL(ST, BNT_, ACC)                                     \
    igemm_bwd_data_k<T, ST, BNT_, ACC><<<grid, TPB, 0, s>>>(              \
        (const T*)dy, (const T*)w2, (T*)dx, (const T*)zero_page, Nn, H,   \
        W, C, P, Q, Kout, R, S, stride, pad, tiles_n, (const T*)accum)
  PDTA_DISPATCH_DTYPE16(dt, T, {
    if (bnt == 128) {
      if (stride == 1) {
        if (accum) PDTA_BWD_SMALL(1, 128, true);
        else PDTA_BWD_SMALL(1, 128, false);
      } else if (stride == 2) {
        if (accum) PDTA_BWD_SMALL(2, 128, true);
        else PDTA_BWD_SMALL(2, 128, false);
      } else {
        if (accum) PDTA_BWD_SMALL(0, 128, true);
        else PDTA_BWD_SMALL(0, 128, false);
      }
    } else {
      if (stride == 1) {
        if (accum) PDTA_BWD_SMALL(1, 64, true);
        else PDTA_BWD_SMALL(1, 64, false);
      } else if (stride == 2) {
        if (accum) PDTA_BWD_SMALL(2, 64, true);
        else PDTA_BWD_SMALL(2, 64, false);
      } else {
        if (accum) PDTA_BWD_SMALL(0, 64, true);
        else PDTA_BWD_SMALL(0, 64, false);
      }
    }
  });
#undef PDTA_BWD_SMALL
}


int pdta_wrw_use_big(int Nn, int P, int Q, int Kout, int C, int R, int S) {
  if (Kout % 256 != 0 || C % 256 != 0) return 0;
  long NPQ = (long)Nn * P * Q;
  long big_tiles = (long)(Kout / 256) * (C / 256) * R * S;
  return (NPQ >= 16384 && big_tiles >= 16) ? 1 : 0;
}

void pdta_wrw_plan(int Nn, int P, int Q, int Kout, int C, int R, int S,
                   long* tiles_out, long* chunks_out) {
  long NPQ = (long)Nn * P * Q;
  int ko_tiles = (Kout + 63) / 64, c_tiles = (C + 63) / 64;
  int tiles = ko_tiles * c_tiles * R * S;
  long splitk;
  if (pdta_wrw_use_big(Nn, P, Q, Kout, C, R, S)) {
    // 512-thread big-tile blocks: fewer of them, each must carry a long
    // reduction (tiny chunks made the epilogue slab dominate)
    long big_tiles = (long)(ko_tiles / 4) * (c_tiles / 4) * R * S;
    splitk = 1024 / big_tiles;
  } else {
    splitk = 4096 / tiles;
  }
  if (splitk < 1) splitk = 1;
  if (splitk > 256) splitk = 256;
  long rows_per_chunk = ((NPQ + splitk - 1) / splitk + 63) / 64 * 64;
  long chunks = (NPQ + rows_per_chunk - 1) / rows_per_chunk;
  *tiles_out = tiles;
  *chunks_out = chunks;
}

// dw fp32 [Kout,C,R,S] ZEROED by caller; part fp32 [(chunks+16)*tiles*4096]
// (16 extra rows: wrw_mid_k scratch).
void pdta_conv_bwd_weight(const void* x, const void* dy, float* dw,
                          float* part, const void* zero_page, int Nn, int H,
                          int W, int C, int P, int Q, int Kout, int R, int S,
                          int stride, int pad, int dt, hipStream_t s) {
  long NPQ = (long)Nn * P * Q;
  int ko_tiles = (Kout + 63) / 64, c_tiles = (C + 63) / 64;
  long tiles, chunks;
  pdta_wrw_plan(Nn, P, Q, Kout, C, R, S, &tiles, &chunks);
  long rows_per_chunk = ((NPQ + chunks - 1) / chunks + 63) / 64 * 64;
  chunks = (NPQ + rows_per_chunk - 1) / rows_per_chunk;
  bool identity = (R == 1 && S == 1 && stride == 1 && pad == 0);
  if (pdta_wrw_use_big(Nn, P, Q, Kout, C, R, S)) {
    long big_tiles = (long)(ko_tiles / 4) * (c_tiles / 4) * R * S;
    dim3 grid((unsigned)big_tiles, (unsigned)chunks);
    MagicDiv mq = magic_div(Q), mpq = magic_div(P * Q);
    PDTA_DISPATCH_DTYPE16(dt, T, {
      if (identity)
        igemm_wrw_big_k<T, false><<<grid, 512, 0, s>>>(
            (const T*)x, (const T*)dy, part, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, rows_per_chunk, ko_tiles,
            c_tiles, mq.mul, mq.shift, mpq.mul, mpq.shift);
      else
        igemm_wrw_big_k<T, true><<<grid, 512, 0, s>>>(
            (const T*)x, (const T*)dy, part, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, rows_per_chunk, ko_tiles,
            c_tiles, mq.mul, mq.shift, mpq.mul, mpq.shift);
    });
  } else if (Kout % 128 == 0 && C % 128 == 0) {
    // 128x128 tile: half the re-staged dy/x bytes of the 64x64 kernel
    long blocks = (long)R * S * (ko_tiles / 2) * (c_tiles / 2);
    dim3 grid((unsigned)blocks, (unsigned)chunks);
    PDTA_DISPATCH_DTYPE16(dt, T, {
      if (identity)
        igemm_wrw_mid_k<T, false><<<grid, TPB, 0, s>>>(
            (const T*)x, (const T*)dy, part, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, rows_per_chunk, ko_tiles,
            c_tiles, tiles);
      else
        igemm_wrw_mid_k<T, true><<<grid, TPB, 0, s>>>(
            (const T*)x, (const T*)dy, part, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, rows_per_chunk, ko_tiles,
            c_tiles, tiles);
    });
  } else {
    dim3 grid((unsigned)tiles, (unsigned)chunks);
    PDTA_DISPATCH_DTYPE16(dt, T, {
      if (identity)
        igemm_wrw_k<T, false><<<grid, TPB, 0, s>>>(
            (const T*)x, (const T*)dy, part, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, rows_per_chunk, ko_tiles,
            c_tiles);
      else
        igemm_wrw_k<T, true><<<grid, TPB, 0, s>>>(
            (const T*)x, (const T*)dy, part, (const T*)zero_page, Nn, H, W,
            C, P, Q, Kout, R, S, stride, pad, rows_per_chunk, ko_tiles,
            c_tiles);
    });
  }
  long total4 = tiles * 1024;  // float4 elements
  const float* red_src = part;
  int red_chunks = (int)chunks;
  if (chunks > 32) {  // two-level: parallel slab pass, then 16-deep final
    float* mid = part + (long)chunks * tiles * 4096;
    dim3 gmid(ew_grid(total4, 256, 1), 16);
    wrw_mid_k<<<gmid, 256, 0, s>>>(part, mid, (int)chunks, total4);
    red_src = mid;
    red_chunks = 16;
  }
  wrw_reduce_k<<<ew_grid(total4, 256, 1), 256, 0, s>>>(
      red_src, dw, red_chunks, (int)tiles, ko_tiles, c_tiles, C, Kout, R,
      S);
}

void pdta_stem_im2col(const void* x, void* out, int Nn, int H, int W,
                      int Cin, int P, int Q, int R, int S, int stride,
                      int pad, int rsc_pad, int dt, hipStream_t s) {
  // v2 (contiguous per-tap runs, one thread per (row, tap)) measured
  // 656 -> 911 us: per-LANE scalar runs do not coalesce across lanes,
  // losing to v1's Vec8 coalesced writes — v1 stays the shipped path
  long total = (long)Nn * P * Q * (rsc_pad / 8);
  MagicDiv mq = magic_div(Q), mpq = magic_div(P * Q);
  MagicDiv mc = magic_div(Cin), msc = magic_div(S);
  PDTA_DISPATCH_DTYPE16(dt, T, {
    stem_im2col_k<T><<<ew_grid(total, TPB, 1), TPB, 0, s>>>(
        (const T*)x, (T*)out, Nn, H, W, Cin, P, Q, R, S, stride, pad,
        rsc_pad, mq.mul, mq.shift, mpq.mul, mpq.shift, mc.mul, mc.shift,
        msc.mul, msc.shift);
  });
}

}  // extern "C"
