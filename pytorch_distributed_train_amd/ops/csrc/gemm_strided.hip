// Strided-batched GEMM with per-operand transpose handling — the
// attention workhorse (SURVEY.md §2d ViT attention row). Eliminates every
// permute/contiguous copy in multi-head attention (30% of the ViT step):
// operands are read IN PLACE from the interleaved qkv projection
// [b, s, 3, h, hd] via row strides (lda/ldb) + 2-level batch strides
// (batch = (b, h)), and outputs can scatter straight into [b, s, h*hd]
// or d(qkv) layouts via ldc.
//
//   C[m][n] = sum_k A' [m][k] * B' [k][n]
//   TRA=false: A is [m][k] row-major (k contiguous, stride lda) -> normal
//              MFMA fragments; TRA=true: A is [k][m] row-major (m contig,
//              k rows of stride lda) -> ds_read_b64_tr_b16 fragments.
//   TRB mirrors for B ([n][k] vs [k][n]).
// Geometry: 64x64 output tile per block, 4 waves as 2x2 of 32x32
// (the conv-wrw core generalized). K % 64 == 0.
#include "common.h"
#include "mfma_tile.h"

namespace {

constexpr int TPB = 256;

template <typename T, typename OutT, bool TRA, bool TRB>
__global__ __launch_bounds__(TPB) void gemm_bs_k(
    const T* __restrict__ A, const T* __restrict__ B, OutT* __restrict__ C,
    int M, int N, int K, int lda, int ldb, int ldc, int batH, long saB,
    long saH, long sbB, long sbH, long scB, long scH, int tiles_n,
    int karmax, int kbrmax) {
  __shared__ __attribute__((aligned(16))) char lds[2][2][64 * 64 * 2];

  int bb = blockIdx.y / batH, hh = blockIdx.y % batH;
  A += bb * saB + hh * saH;
  B += bb * sbB + hh * sbH;
  C += bb * scB + hh * scH;

  int bid = blockIdx.x;
  int tm = bid / tiles_n, tn = bid % tiles_n;
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int wm = wave >> 1, wn = wave & 1;
  int m0 = tm * 64, n0 = tn * 64;

  int row_in_chunk = lane >> 3;
  int col16 = lane & 7;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // stage one 64x64 operand tile (8 KiB): 2 glds per wave; source carries
  // the chunk^(row&7) swizzle (rule 21). For a normal operand rows are
  // m/n (clamped); for a TR operand rows are k (K%64==0, no clamp).
  auto stage = [&](const T* src, int ld, int roff, int rmax, int coff,
                   char* dst) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int ldsrow = wave * 16 + i * 8 + row_in_chunk;
      int scol16 = col16 ^ (ldsrow & 7);
      int row = roff + ldsrow;
      if (row > rmax) row = rmax;
      const char* gptr = reinterpret_cast<const char*>(src) +
                         ((long)row * ld + coff) * 2 + scol16 * 16;
      glds16(gptr, dst + (wave * 16 + i * 8) * 128);
    }
  };
  // karmax/kbrmax: last readable k-row of a TR operand (normally K-1).
  // An UNPADDED operand (attention's Q/dOut when qkv is packed at 197
  // rows) passes its real last row; the clamped re-reads contribute
  // garbage that the other operand's exactly-zero tail rows annihilate.
  auto stage_tiles = [&](int k0, int buf) {
    if (TRA)
      stage(A, lda, k0, karmax, m0, lds[buf][0]);
    else
      stage(A, lda, m0, M - 1, k0, lds[buf][0]);
    if (TRB)
      stage(B, ldb, k0, kbrmax, n0, lds[buf][1]);
    else
      stage(B, ldb, n0, N - 1, k0, lds[buf][1]);
  };

  stage_tiles(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int kt = 0; kt < K / 64; ++kt) {
    if (kt + 1 < K / 64) stage_tiles((kt + 1) * 64, cur ^ 1);
    unsigned la = lds_addr_of(lds[cur][0]);
    unsigned lb = lds_addr_of(lds[cur][1]);
    const char* la_p = lds[cur][0];
    const char* lb_p = lds[cur][1];
    // issue all LDS reads for both K=32 steps, wait once, then MFMA
    unsigned long long alo[2][2], ahi[2][2], blo[2][2], bhi[2][2];
    s16x8 afn[2][2], bfn[2][2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        if (TRA)
          frag_read_tr_issue(la, kk * 32, wm * 32 + i * 16, lane,
                             &alo[kk][i], &ahi[kk][i]);
        else
          afn[kk][i] = frag_read128(la_p, wm * 32 + i * 16 + (lane & 15),
                                    (lane >> 4) + kk * 4);
        if (TRB)
          frag_read_tr_issue(lb, kk * 32, wn * 32 + i * 16, lane,
                             &blo[kk][i], &bhi[kk][i]);
        else
          bfn[kk][i] = frag_read128(lb_p, wn * 32 + i * 16 + (lane & 15),
                                    (lane >> 4) + kk * 4);
      }
    }
    if (TRA || TRB) tr_read_wait();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      s16x8 af[2], bf[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        af[i] = TRA ? tr_pair_to_frag(alo[kk][i], ahi[kk][i]) : afn[kk][i];
        bf[i] = TRB ? tr_pair_to_frag(blo[kk][i], bhi[kk][i]) : bfn[kk][i];
      }
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = MfmaOp<T>::run(af[i], bf[j], acc[i][j]);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int rbase = m0 + wm * 32 + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int col = n0 + wn * 32 + j * 16 + (lane & 15);
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = rbase + r;
        if (row < M) C[(long)row * ldc + col] = from_f32<OutT>(acc[i][j][r]);
      }
    }
  }
}

}  // namespace

extern "C" {

void pdta_gemm_bs(const void* A, const void* B, void* C, int M, int N,
                  int K, int lda, int ldb, int ldc, int batches, int batH,
                  long saB, long saH, long sbB, long sbH, long scB,
                  long scH, int tra, int trb, int out_f32, int karmax,
                  int kbrmax, int dt, hipStream_t s) {
  int tiles_m = (M + 63) / 64, tiles_n = (N + 63) / 64;
  dim3 grid(tiles_m * tiles_n, batches);
#define LAUNCH(TA, TB, OT)                                               \
  gemm_bs_k<T, OT, TA, TB><<<grid, TPB, 0, s>>>(                         \
      (const T*)A, (const T*)B, (OT*)C, M, N, K, lda, ldb, ldc, batH,    \
      saB, saH, sbB, sbH, scB, scH, tiles_n, karmax, kbrmax)
  PDTA_DISPATCH_DTYPE16(dt, T, {
    if (out_f32) {
      if (tra && trb) LAUNCH(true, true, float);
      else if (tra) LAUNCH(true, false, float);
      else if (trb) LAUNCH(false, true, float);
      else LAUNCH(false, false, float);
    } else {
      if (tra && trb) LAUNCH(true, true, T);
      else if (tra) LAUNCH(true, false, T);
      else if (trb) LAUNCH(false, true, T);
      else LAUNCH(false, false, T);
    }
  });
#undef LAUNCH
}

}  // extern "C"
