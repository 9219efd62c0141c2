// BatchNorm kernels over [N,H,W,C] (C innermost, C % 64 == 0).
// Rebuilds torch's batch_norm_stats / batch_norm_elemt /
// batch_norm_backward_reduce / batch_norm_backward_elemt set
// (SURVEY.md §2d BatchNorm row) as NHWC-native CDNA4 kernels.
//
// Reductions are TWO-STAGE with plain stores (no global atomics, no
// pre-zeroed buffers — at B=256 the per-layer at::zeros launches were ~3%
// of the step): stage 1 writes per-chunk partials [chunk][2C] (each thread
// owns one channel column, strides rows — coalesced since consecutive
// lanes read consecutive channels; LDS combine across the row-parallel
// threads); stage 2 reduces chunks, computes mean/invstd, and (training
// path) updates the running stats in the same kernel.
//
// The ReLU that precedes BN backward is fused: when `relu_y` is non-null,
// dy is masked by (y > 0) on the fly in both backward kernels, removing
// the separate relu_bwd_mask pass over dy.
#include "common.h"

namespace {

constexpr int TPB = 256;

// -- stage 1: per-chunk partial sums (sum, sumsq) -------------------------
template <typename T>
__global__ void bn_part_k(const T* __restrict__ x, float* __restrict__ part,
                          long rows, int C, int rows_per_block) {
  __shared__ float red[2][TPB];
  int cpb = min(C, TPB);
  int rpar = TPB / cpb;
  int lane_c = (int)(threadIdx.x % cpb);
  int c = blockIdx.x * cpb + lane_c;
  long r0 = (long)blockIdx.y * rows_per_block + threadIdx.x / cpb;
  long r_end = min(rows, (long)(blockIdx.y + 1) * rows_per_block);
  float s = 0.f, ss = 0.f;
  for (long r = r0; r < r_end; r += rpar) {
    float v = to_f32(x[r * C + c]);
    s += v;
    ss += v * v;
  }
  // combine the rpar threads sharing this channel
  red[0][threadIdx.x] = s;
  red[1][threadIdx.x] = ss;
  __syncthreads();
  if (threadIdx.x < cpb) {
    for (int i = 1; i < rpar; ++i) {
      s += red[0][threadIdx.x + i * cpb];
      ss += red[1][threadIdx.x + i * cpb];
    }
    float* p = part + ((long)blockIdx.y * C + blockIdx.x * cpb) * 2;
    p[lane_c] = s;
    p[cpb + lane_c] = ss;
  }
}

// -- stage 2: reduce chunks -> mean/var (+invstd, +running update) --------
// finalize: block = 64 channels x 4 chunk-slices (full waves even at
// C=64; the slices LDS-reduce) — a C-threads-only version was one wave on
// the whole chip and chunk-serial (350 us; this form is ~5 us).
__global__ void bn_stats_final_k(const float* __restrict__ part, int chunks,
                                 int C, float inv_n, float unbias,
                                 float* __restrict__ mean,
                                 float* __restrict__ var_or_invstd,
                                 float* __restrict__ running_mean,
                                 float* __restrict__ running_var,
                                 float momentum, float eps, int mode) {
  // mode 0: write var (SyncBN partials path). mode 1: write invstd and
  // update running stats (plain BN training fast path).
  __shared__ float red[2][TPB];
  int c = blockIdx.x * 64 + (int)(threadIdx.x % 64);
  int slice = threadIdx.x / 64;  // 4 slices
  int cpb = min(C, TPB);
  int cblk = c / cpb, lane_c = c % cpb;
  float s = 0.f, ss = 0.f;
  if (c < C) {
    for (int k = slice; k < chunks; k += 4) {
      const float* p = part + ((long)k * C + cblk * cpb) * 2;
      s += p[lane_c];
      ss += p[cpb + lane_c];
    }
  }
  red[0][threadIdx.x] = s;
  red[1][threadIdx.x] = ss;
  __syncthreads();
  if (slice == 0 && c < C) {
#pragma unroll
    for (int i = 1; i < 4; ++i) {
      s += red[0][threadIdx.x + i * 64];
      ss += red[1][threadIdx.x + i * 64];
    }
    float m = s * inv_n;
    float v = fmaxf(ss * inv_n - m * m, 0.f);
    mean[c] = m;
    if (mode == 0) {
      var_or_invstd[c] = v;
    } else {
      var_or_invstd[c] = rsqrtf(v + eps);
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * m;
      running_var[c] =
          (1.f - momentum) * running_var[c] + momentum * v * unbias;
    }
  }
}

// -- y = (x - mean) * invstd * w + b, optional ReLU ----------------------
template <typename T, bool RELU, bool RES>
__global__ void bn_apply_k(const T* __restrict__ x, T* __restrict__ y,
                           const T* __restrict__ res,
                           const float* __restrict__ weight,
                           const float* __restrict__ bias,
                           const float* __restrict__ mean,
                           const float* __restrict__ invstd, long rows,
                           int c8,  // c8 = C/8
                           unsigned char* __restrict__ mask_out = nullptr) {
  // mask_out (joins only): 1 bit per element, bit j of byte (r*C+c0)/8 =
  // (rounded output > 0) — the backward reduce reads 1/16 the bytes of
  // the stored relu output it replaces
  int cpg = min(c8, TPB);
  int rpar = TPB / cpg;
  int cg = blockIdx.x * cpg + (int)(threadIdx.x % cpg);
  if (cg >= c8) return;
  int c0 = cg * 8;
  float scale[8], shift[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float is = invstd[c0 + j];
    scale[j] = weight[c0 + j] * is;
    shift[j] = bias[c0 + j] - mean[c0 + j] * scale[j];
  }
  const int C = c8 * 8;
  long rstride = (long)gridDim.y * rpar;
  long r = (long)blockIdx.y * rpar + threadIdx.x / cpg;
  for (; r + rstride < rows; r += 2 * rstride) {  // 2-4 loads in flight
    long r2 = r + rstride;
    Vec8<T> va = load8(x + r * C + c0);
    Vec8<T> vb = load8(x + r2 * C + c0);
    Vec8<T> ra, rb;
    if (RES) {
      ra = load8(res + r * C + c0);
      rb = load8(res + r2 * C + c0);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float o0 = to_f32(va.v[j]) * scale[j] + shift[j];
      float o1 = to_f32(vb.v[j]) * scale[j] + shift[j];
      if (RES) {
        o0 += to_f32(ra.v[j]);
        o1 += to_f32(rb.v[j]);
      }
      if (RELU) {
        o0 = fmaxf(o0, 0.f);
        o1 = fmaxf(o1, 0.f);
      }
      va.v[j] = from_f32<T>(o0);
      vb.v[j] = from_f32<T>(o1);
    }
    if (RELU && RES && mask_out) {
      unsigned char m0 = 0, m1 = 0;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        m0 |= (to_f32(va.v[j]) > 0.f) << j;
        m1 |= (to_f32(vb.v[j]) > 0.f) << j;
      }
      mask_out[(r * C + c0) >> 3] = m0;
      mask_out[(r2 * C + c0) >> 3] = m1;
    }
    store8(y + r * C + c0, va);
    store8(y + r2 * C + c0, vb);
  }
  for (; r < rows; r += rstride) {
    Vec8<T> v = load8(x + r * C + c0);
    Vec8<T> rv;
    if (RES) rv = load8(res + r * C + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float o = to_f32(v.v[j]) * scale[j] + shift[j];
      if (RES) o += to_f32(rv.v[j]);
      if (RELU) o = fmaxf(o, 0.f);
      v.v[j] = from_f32<T>(o);
    }
    if (RELU && RES && mask_out) {
      unsigned char m = 0;
#pragma unroll
      for (int j = 0; j < 8; ++j) m |= (to_f32(v.v[j]) > 0.f) << j;
      mask_out[(r * C + c0) >> 3] = m;
    }
    store8(y + r * C + c0, v);
  }
}

// -- backward stage 1: per-chunk partial (sum_dy, sum_dy_xmu), with the
//    preceding-ReLU mask fused. MASK 0: none. MASK 1: dy := y>0 ? dy : 0
//    (read the stored relu output — required for residual joins, where the
//    pre-activation includes the identity branch). MASK 2: recompute the
//    mask as gamma*xhat+beta > 0 from x — one whole HBM pass over relu_y
//    never happens (exact up to bf16 boundary rounding of y).
//    MASK 3: 1-bit mask emitted by the join's bn_apply (mbits) — same
//    semantics as MASK 1 at 1/16 the mask bytes ------------------------
template <typename T, int MASK, bool DY2 = false>
__global__ void bn_bwd_part_k(const T* __restrict__ x,
                              const T* __restrict__ dy,
                              const T* __restrict__ relu_y,
                              const float* __restrict__ mean,
                              const float* __restrict__ invstd,
                              const float* __restrict__ weight,
                              const float* __restrict__ bias,
                              float* __restrict__ part, long rows, int C,
                              int rows_per_block,
                              const T* __restrict__ dy2 = nullptr,
                              T* __restrict__ dres_out = nullptr,
                              const unsigned char* __restrict__ mbits =
                                  nullptr) {
  // DY2 also WRITES the masked combined dy (== the residual-path
  // gradient dres): the elemt pass then reads that single tensor with no
  // mask and no dres store — 8 HBM passes for the whole join backward
  // instead of 11 with an autograd fork add
  // DY2: second incoming gradient (residual fork) summed on load — the
  // whole-tensor autograd add this replaces was 3 HBM passes; reading
  // dy2 here and in the elemt kernel is 2
  __shared__ float red[2][TPB];
  int cpb = min(C, TPB);
  int rpar = TPB / cpb;
  int lane_c = (int)(threadIdx.x % cpb);
  int c = blockIdx.x * cpb + lane_c;
  long r0 = (long)blockIdx.y * rows_per_block + threadIdx.x / cpb;
  long r_end = min(rows, (long)(blockIdx.y + 1) * rows_per_block);
  float m = mean[c];
  float k1 = 0.f, b0 = 0.f;
  if (MASK == 2) {
    k1 = weight[c] * invstd[c];
    b0 = bias[c];
  }
  float s = 0.f, sx = 0.f;
  // 4 rows per iteration: keeps 8-12 loads in flight per thread (the
  // rolled loop was ~2x off the traffic floor, latency-bound)
  long r = r0;
  // DY2: 8 rows per iteration — the 4-stream two-dy variant was
  // ~2x off its traffic floor at 4 rows in flight
  if (DY2)
    for (; r + 7 * rpar < r_end; r += 8 * rpar) {
    float g0 = to_f32(dy[r * C + c]);
    float g1 = to_f32(dy[(r + rpar) * C + c]);
    float g2 = to_f32(dy[(r + 2 * rpar) * C + c]);
    float g3 = to_f32(dy[(r + 3 * rpar) * C + c]);
    float h0, h1, h2, h3;
    if constexpr (DY2) {
      // issue all four dy2 loads before any arithmetic consumes them
      h0 = to_f32(dy2[r * C + c]);
      h1 = to_f32(dy2[(r + rpar) * C + c]);
      h2 = to_f32(dy2[(r + 2 * rpar) * C + c]);
      h3 = to_f32(dy2[(r + 3 * rpar) * C + c]);
    }
    float v0 = to_f32(x[r * C + c]);
    float v1 = to_f32(x[(r + rpar) * C + c]);
    float v2 = to_f32(x[(r + 2 * rpar) * C + c]);
    float v3 = to_f32(x[(r + 3 * rpar) * C + c]);
    if constexpr (DY2) {
      g0 += h0;
      g1 += h1;
      g2 += h2;
      g3 += h3;
    }
    if (MASK == 1) {
      g0 = to_f32(relu_y[r * C + c]) > 0.f ? g0 : 0.f;
      g1 = to_f32(relu_y[(r + rpar) * C + c]) > 0.f ? g1 : 0.f;
      g2 = to_f32(relu_y[(r + 2 * rpar) * C + c]) > 0.f ? g2 : 0.f;
      g3 = to_f32(relu_y[(r + 3 * rpar) * C + c]) > 0.f ? g3 : 0.f;
    }
    if (MASK == 3) {
      int bit = c & 7;
      g0 = (mbits[(r * C + c) >> 3] >> bit) & 1 ? g0 : 0.f;
      g1 = (mbits[((r + rpar) * C + c) >> 3] >> bit) & 1 ? g1 : 0.f;
      g2 = (mbits[((r + 2 * rpar) * C + c) >> 3] >> bit) & 1 ? g2 : 0.f;
      g3 = (mbits[((r + 3 * rpar) * C + c) >> 3] >> bit) & 1 ? g3 : 0.f;
    }
    if (MASK == 2) {
      g0 = k1 * (v0 - m) + b0 > 0.f ? g0 : 0.f;
      g1 = k1 * (v1 - m) + b0 > 0.f ? g1 : 0.f;
      g2 = k1 * (v2 - m) + b0 > 0.f ? g2 : 0.f;
      g3 = k1 * (v3 - m) + b0 > 0.f ? g3 : 0.f;
    }
    if constexpr (DY2) {
      dres_out[r * C + c] = from_f32<T>(g0);
      dres_out[(r + rpar) * C + c] = from_f32<T>(g1);
      dres_out[(r + 2 * rpar) * C + c] = from_f32<T>(g2);
      dres_out[(r + 3 * rpar) * C + c] = from_f32<T>(g3);
    }
    s += g0 + g1 + g2 + g3;
    sx += g0 * (v0 - m) + g1 * (v1 - m) + g2 * (v2 - m) + g3 * (v3 - m);
      float g0b = to_f32(dy[(r + 4 * rpar) * C + c]);
    float g1b = to_f32(dy[(r + 5 * rpar) * C + c]);
    float g2b = to_f32(dy[(r + 6 * rpar) * C + c]);
    float g3b = to_f32(dy[(r + 7 * rpar) * C + c]);
    float h0b, h1b, h2b, h3b;
    if constexpr (DY2) {
      // issue all four dy2 loads before any arithmetic consumes them
      h0b = to_f32(dy2[(r + 4 * rpar) * C + c]);
      h1b = to_f32(dy2[(r + 5 * rpar) * C + c]);
      h2b = to_f32(dy2[(r + 6 * rpar) * C + c]);
      h3b = to_f32(dy2[(r + 7 * rpar) * C + c]);
    }
    float v0b = to_f32(x[(r + 4 * rpar) * C + c]);
    float v1b = to_f32(x[(r + 5 * rpar) * C + c]);
    float v2b = to_f32(x[(r + 6 * rpar) * C + c]);
    float v3b = to_f32(x[(r + 7 * rpar) * C + c]);
    if constexpr (DY2) {
      g0b += h0b;
      g1b += h1b;
      g2b += h2b;
      g3b += h3b;
    }
    if (MASK == 1) {
      g0b = to_f32(relu_y[(r + 4 * rpar) * C + c]) > 0.f ? g0b : 0.f;
      g1b = to_f32(relu_y[(r + 5 * rpar) * C + c]) > 0.f ? g1b : 0.f;
      g2b = to_f32(relu_y[(r + 6 * rpar) * C + c]) > 0.f ? g2b : 0.f;
      g3b = to_f32(relu_y[(r + 7 * rpar) * C + c]) > 0.f ? g3b : 0.f;
    }
    if (MASK == 3) {
      int bit = c & 7;
      g0b = (mbits[((r + 4 * rpar) * C + c) >> 3] >> bit) & 1 ? g0b : 0.f;
      g1b = (mbits[((r + 5 * rpar) * C + c) >> 3] >> bit) & 1 ? g1b : 0.f;
      g2b = (mbits[((r + 6 * rpar) * C + c) >> 3] >> bit) & 1 ? g2b : 0.f;
      g3b = (mbits[((r + 7 * rpar) * C + c) >> 3] >> bit) & 1 ? g3b : 0.f;
    }
    if (MASK == 2) {
      g0b = k1 * (v0b - m) + b0 > 0.f ? g0b : 0.f;
      g1b = k1 * (v1b - m) + b0 > 0.f ? g1b : 0.f;
      g2b = k1 * (v2b - m) + b0 > 0.f ? g2b : 0.f;
      g3b = k1 * (v3b - m) + b0 > 0.f ? g3b : 0.f;
    }
    if constexpr (DY2) {
      dres_out[(r + 4 * rpar) * C + c] = from_f32<T>(g0b);
      dres_out[(r + 5 * rpar) * C + c] = from_f32<T>(g1b);
      dres_out[(r + 6 * rpar) * C + c] = from_f32<T>(g2b);
      dres_out[(r + 7 * rpar) * C + c] = from_f32<T>(g3b);
    }
    s += g0b + g1b + g2b + g3b;
    sx += g0b * (v0b - m) + g1b * (v1b - m) + g2b * (v2b - m) + g3b * (v3b - m);
      }
  for (; r + 3 * rpar < r_end; r += 4 * rpar) {
    float g0 = to_f32(dy[r * C + c]);
    float g1 = to_f32(dy[(r + rpar) * C + c]);
    float g2 = to_f32(dy[(r + 2 * rpar) * C + c]);
    float g3 = to_f32(dy[(r + 3 * rpar) * C + c]);
    float h0, h1, h2, h3;
    if constexpr (DY2) {
      // issue all four dy2 loads before any arithmetic consumes them
      h0 = to_f32(dy2[r * C + c]);
      h1 = to_f32(dy2[(r + rpar) * C + c]);
      h2 = to_f32(dy2[(r + 2 * rpar) * C + c]);
      h3 = to_f32(dy2[(r + 3 * rpar) * C + c]);
    }
    float v0 = to_f32(x[r * C + c]);
    float v1 = to_f32(x[(r + rpar) * C + c]);
    float v2 = to_f32(x[(r + 2 * rpar) * C + c]);
    float v3 = to_f32(x[(r + 3 * rpar) * C + c]);
    if constexpr (DY2) {
      g0 += h0;
      g1 += h1;
      g2 += h2;
      g3 += h3;
    }
    if (MASK == 1) {
      g0 = to_f32(relu_y[r * C + c]) > 0.f ? g0 : 0.f;
      g1 = to_f32(relu_y[(r + rpar) * C + c]) > 0.f ? g1 : 0.f;
      g2 = to_f32(relu_y[(r + 2 * rpar) * C + c]) > 0.f ? g2 : 0.f;
      g3 = to_f32(relu_y[(r + 3 * rpar) * C + c]) > 0.f ? g3 : 0.f;
    }
    if (MASK == 3) {
      int bit = c & 7;
      g0 = (mbits[(r * C + c) >> 3] >> bit) & 1 ? g0 : 0.f;
      g1 = (mbits[((r + rpar) * C + c) >> 3] >> bit) & 1 ? g1 : 0.f;
      g2 = (mbits[((r + 2 * rpar) * C + c) >> 3] >> bit) & 1 ? g2 : 0.f;
      g3 = (mbits[((r + 3 * rpar) * C + c) >> 3] >> bit) & 1 ? g3 : 0.f;
    }
    if (MASK == 2) {
      g0 = k1 * (v0 - m) + b0 > 0.f ? g0 : 0.f;
      g1 = k1 * (v1 - m) + b0 > 0.f ? g1 : 0.f;
      g2 = k1 * (v2 - m) + b0 > 0.f ? g2 : 0.f;
      g3 = k1 * (v3 - m) + b0 > 0.f ? g3 : 0.f;
    }
    if constexpr (DY2) {
      dres_out[r * C + c] = from_f32<T>(g0);
      dres_out[(r + rpar) * C + c] = from_f32<T>(g1);
      dres_out[(r + 2 * rpar) * C + c] = from_f32<T>(g2);
      dres_out[(r + 3 * rpar) * C + c] = from_f32<T>(g3);
    }
    s += g0 + g1 + g2 + g3;
    sx += g0 * (v0 - m) + g1 * (v1 - m) + g2 * (v2 - m) + g3 * (v3 - m);
  }
  for (; r < r_end; r += rpar) {
    float g = to_f32(dy[r * C + c]);
    if constexpr (DY2) g += to_f32(dy2[r * C + c]);
    float v = to_f32(x[r * C + c]);
    if (MASK == 1) g = to_f32(relu_y[r * C + c]) > 0.f ? g : 0.f;
    if (MASK == 2) g = k1 * (v - m) + b0 > 0.f ? g : 0.f;
    if (MASK == 3) g = (mbits[(r * C + c) >> 3] >> (c & 7)) & 1 ? g : 0.f;
    if constexpr (DY2) dres_out[r * C + c] = from_f32<T>(g);
    s += g;
    sx += g * (v - m);
  }
  red[0][threadIdx.x] = s;
  red[1][threadIdx.x] = sx;
  __syncthreads();
  if (threadIdx.x < cpb) {
    for (int i = 1; i < rpar; ++i) {
      s += red[0][threadIdx.x + i * cpb];
      sx += red[1][threadIdx.x + i * cpb];
    }
    float* p = part + ((long)blockIdx.y * C + blockIdx.x * cpb) * 2;
    p[lane_c] = s;
    p[cpb + lane_c] = sx;
  }
}

__global__ void bn_bwd_final_k(const float* __restrict__ part, int chunks,
                               int C, float* __restrict__ sum_dy,
                               float* __restrict__ sum_dy_xmu) {
  __shared__ float red[2][TPB];
  int c = blockIdx.x * 64 + (int)(threadIdx.x % 64);
  int slice = threadIdx.x / 64;
  int cpb = min(C, TPB);
  int cblk = c / cpb, lane_c = c % cpb;
  float s = 0.f, sx = 0.f;
  if (c < C) {
    for (int k = slice; k < chunks; k += 4) {
      const float* p = part + ((long)k * C + cblk * cpb) * 2;
      s += p[lane_c];
      sx += p[cpb + lane_c];
    }
  }
  red[0][threadIdx.x] = s;
  red[1][threadIdx.x] = sx;
  __syncthreads();
  if (slice == 0 && c < C) {
#pragma unroll
    for (int i = 1; i < 4; ++i) {
      s += red[0][threadIdx.x + i * 64];
      sx += red[1][threadIdx.x + i * 64];
    }
    sum_dy[c] = s;
    sum_dy_xmu[c] = sx;
  }
}

// -- backward elementwise (ReLU mask fused when RELU):
//    dx = w*invstd * (dy - sum_dy/n - (x-mean)*invstd^2*sum_dy_xmu/n)
template <typename T, int MASK, bool DRES, bool DY2 = false>
__global__ void bn_bwd_elemt_k(const T* __restrict__ x,
                               const T* __restrict__ dy,
                               const T* __restrict__ relu_y,
                               T* __restrict__ dx, T* __restrict__ dres,
                               const float* __restrict__ weight,
                               const float* __restrict__ bias,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               const float* __restrict__ sum_dy,
                               const float* __restrict__ sum_dy_xmu,
                               float inv_n, long rows, int c8,
                               const T* __restrict__ dy2 = nullptr) {
  int cpg = min(c8, TPB);
  int rpar = TPB / cpg;
  int cg = blockIdx.x * cpg + (int)(threadIdx.x % cpg);
  if (cg >= c8) return;
  int c0 = cg * 8;
  float k1[8], k2[8], k3[8], mu[8], b0[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float is = invstd[c0 + j];
    k1[j] = weight[c0 + j] * is;
    k2[j] = sum_dy[c0 + j] * inv_n;
    k3[j] = is * is * sum_dy_xmu[c0 + j] * inv_n;
    mu[j] = mean[c0 + j];
    b0[j] = (MASK == 2) ? bias[c0 + j] : 0.f;
  }
  const int C = c8 * 8;
  long rstride = (long)gridDim.y * rpar;
  long r = (long)blockIdx.y * rpar + threadIdx.x / cpg;
  // 2 rows per iteration: 4-6 loads in flight (latency hiding)
  for (; r + rstride < rows; r += 2 * rstride) {
    long r2 = r + rstride;
    Vec8<T> ga = load8(dy + r * C + c0);
    Vec8<T> gb = load8(dy + r2 * C + c0);
    Vec8<T> g2a, g2b;
    if constexpr (DY2) {
      g2a = load8(dy2 + r * C + c0);
      g2b = load8(dy2 + r2 * C + c0);
    }
    Vec8<T> va = load8(x + r * C + c0);
    Vec8<T> vb = load8(x + r2 * C + c0);
    Vec8<T> ya, yb, da, db;
    if (MASK == 1) {
      ya = load8(relu_y + r * C + c0);
      yb = load8(relu_y + r2 * C + c0);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g0 = to_f32(ga.v[j]), g1 = to_f32(gb.v[j]);
      if constexpr (DY2) {
        g0 += to_f32(g2a.v[j]);
        g1 += to_f32(g2b.v[j]);
      }
      float xm0 = to_f32(va.v[j]) - mu[j];
      float xm1 = to_f32(vb.v[j]) - mu[j];
      if (MASK == 1) {
        g0 = to_f32(ya.v[j]) > 0.f ? g0 : 0.f;
        g1 = to_f32(yb.v[j]) > 0.f ? g1 : 0.f;
      }
      if (MASK == 2) {
        g0 = k1[j] * xm0 + b0[j] > 0.f ? g0 : 0.f;
        g1 = k1[j] * xm1 + b0[j] > 0.f ? g1 : 0.f;
      }
      if (DRES) {
        da.v[j] = from_f32<T>(g0);
        db.v[j] = from_f32<T>(g1);
      }
      ga.v[j] = from_f32<T>(k1[j] * (g0 - k2[j] - xm0 * k3[j]));
      gb.v[j] = from_f32<T>(k1[j] * (g1 - k2[j] - xm1 * k3[j]));
    }
    if (DRES) {
      store8(dres + r * C + c0, da);
      store8(dres + r2 * C + c0, db);
    }
    store8(dx + r * C + c0, ga);
    store8(dx + r2 * C + c0, gb);
  }
  for (; r < rows; r += rstride) {
    Vec8<T> g = load8(dy + r * C + c0);
    Vec8<T> g2;
    if constexpr (DY2) g2 = load8(dy2 + r * C + c0);
    Vec8<T> v = load8(x + r * C + c0);
    Vec8<T> yv, dr;
    if (MASK == 1) yv = load8(relu_y + r * C + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gg = to_f32(g.v[j]);
      if constexpr (DY2) gg += to_f32(g2.v[j]);
      float xm = to_f32(v.v[j]) - mu[j];
      if (MASK == 1) gg = to_f32(yv.v[j]) > 0.f ? gg : 0.f;
      if (MASK == 2) gg = k1[j] * xm + b0[j] > 0.f ? gg : 0.f;
      if (DRES) dr.v[j] = from_f32<T>(gg);
      float o = k1[j] * (gg - k2[j] - xm * k3[j]);
      g.v[j] = from_f32<T>(o);
    }
    if (DRES) store8(dres + r * C + c0, dr);
    store8(dx + r * C + c0, g);
  }
}


// -- optional middle reduction: [chunks][2C] -> [16][2C] so the finalize
// kernels never loop more than 16 chunks (a 2048-chunk finalize at C=64
// was 95 us; stage1+mid+final is ~15 us total) -----------------------------
__global__ void bn_mid_k(const float* __restrict__ in,
                         float* __restrict__ out, int chunks, int twoC) {
  int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= twoC) return;
  float s = 0.f;
  for (int k = blockIdx.y; k < chunks; k += 64)
    s += in[(long)k * twoC + col];
  out[(long)blockIdx.y * twoC + col] = s;
}

static inline dim3 reduce_grid(long rows, int C, int* rows_per_block,
                               int* chunks) {
  int cpb = C < TPB ? C : TPB;
  int cblocks = (C + cpb - 1) / cpb;
  long want = 2048 / cblocks;
  if (want < 1) want = 1;
  long rpb = (rows + want - 1) / want;
  int rpar = TPB / cpb;
  if (rpb < rpar) rpb = rpar;
  *rows_per_block = (int)rpb;
  long rblocks = (rows + rpb - 1) / rpb;
  *chunks = (int)rblocks;
  return dim3(cblocks, (unsigned)rblocks);
}

static inline dim3 ew_grid2(long rows, int C) {
  int c8 = C / 8;
  int cpg = c8 < TPB ? c8 : TPB;
  int cgb = (c8 + cpg - 1) / cpg;
  int rpar = TPB / cpg;
  long rblocks = (rows + rpar - 1) / rpar;
  long want = 2048 / cgb;
  if (want < 1) want = 1;
  if (rblocks > want) rblocks = want;
  return dim3(cgb, (unsigned)rblocks);
}


// ---- SyncBN pack/combine: the per-BN python op chain
// (mul/add/cat/slice ~12 launches per layer per step) fused into single
// kernels around the packed [sum ‖ sumsq ‖ count] collective.
__global__ void syncbn_pack_k(const float* __restrict__ mean,
                              const float* __restrict__ var, float n,
                              float* __restrict__ out, int C) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < C) {
    float m = mean[i];
    out[i] = m * n;
    out[C + i] = (var[i] + m * m) * n;
  }
  if (i == 0) out[2 * C] = n;
}

// gathered [ws][2C+1] -> global mean + invstd (+ running stats update,
// count-weighted unbiased var) + total count
__global__ void syncbn_combine_k(const float* __restrict__ g, int ws,
                                 int C, float eps, float momentum,
                                 float* __restrict__ mean,
                                 float* __restrict__ invstd,
                                 float* __restrict__ running_mean,
                                 float* __restrict__ running_var,
                                 float* __restrict__ total_out) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= C) return;
  int stride = 2 * C + 1;
  float s = 0.f, ss = 0.f, n = 0.f;
  for (int r = 0; r < ws; ++r) {
    const float* row = g + (long)r * stride;
    s += row[i];
    ss += row[C + i];
    n += row[2 * C];
  }
  float m = s / n;
  float v = fmaxf(ss / n - m * m, 0.f);
  mean[i] = m;
  invstd[i] = rsqrtf(v + eps);
  if (running_mean) {
    float unb = v * (n / fmaxf(n - 1.f, 1.f));
    running_mean[i] = running_mean[i] * (1.f - momentum) + m * momentum;
    running_var[i] = running_var[i] * (1.f - momentum) + unb * momentum;
  }
  if (i == 0 && total_out) *total_out = n;
}

// backward: [sum_dy ‖ sum_dy_xmu ‖ n] pack, and post-allreduce in-place
// normalize (first 2C /= last element)
__global__ void syncbn_pack2_k(const float* __restrict__ a,
                               const float* __restrict__ b, float n,
                               float* __restrict__ out, int C) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < C) {
    out[i] = a[i];
    out[C + i] = b[i];
  }
  if (i == 0) out[2 * C] = n;
}

__global__ void syncbn_div_k(float* __restrict__ p, int C) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < 2 * C) p[i] /= p[2 * C];
}

}  // namespace

extern "C" {

// part: fp32 workspace [chunks][2C] (uninitialized ok). mode 1: training
// fast path (invstd out + running stats update); mode 0: (mean, var) out.
void pdta_bn_stats(const void* x, float* part, float* mean, float* out2,
                   float* running_mean, float* running_var, float momentum,
                   float eps, int mode, long rows, int C, int dt,
                   hipStream_t s) {
  int rpb, chunks;
  dim3 g = reduce_grid(rows, C, &rpb, &chunks);
  PDTA_DISPATCH_DTYPE(dt, T, {
    bn_part_k<T><<<g, TPB, 0, s>>>((const T*)x, part, rows, C, rpb);
  });
  const float* fin = part;
  int fchunks = chunks;
  if (chunks > 64) {
    float* part2 = part + (long)chunks * 2 * C;
    bn_mid_k<<<dim3((2 * C + 255) / 256, 64), 256, 0, s>>>(part, part2,
                                                           chunks, 2 * C);
    fin = part2;
    fchunks = 64;
  }
  float unbias = rows > 1 ? (float)rows / (float)(rows - 1) : 1.f;
  bn_stats_final_k<<<(C + 63) / 64, 256, 0, s>>>(
      fin, fchunks, C, 1.f / (float)rows, unbias, mean, out2, running_mean,
      running_var, momentum, eps, mode);
}

// Finalize-only path: `part` already holds [chunks][2C] partials written by
// a producing kernel's epilogue (conv.hip igemm_fwd STATS variants) — the
// stage-1 read of x never happens. part must carry 64 extra scratch rows.
void pdta_bn_stats_from_part(float* part, int chunks, float* mean,
                             float* out2, float* running_mean,
                             float* running_var, float momentum, float eps,
                             int mode, long rows, int C, hipStream_t s) {
  const float* fin = part;
  int fchunks = chunks;
  if (chunks > 64) {
    float* part2 = part + (long)chunks * 2 * C;
    bn_mid_k<<<dim3((2 * C + 255) / 256, 64), 256, 0, s>>>(part, part2,
                                                           chunks, 2 * C);
    fin = part2;
    fchunks = 64;
  }
  float unbias = rows > 1 ? (float)rows / (float)(rows - 1) : 1.f;
  bn_stats_final_k<<<(C + 63) / 64, 256, 0, s>>>(
      fin, fchunks, C, 1.f / (float)rows, unbias, mean, out2, running_mean,
      running_var, momentum, eps, mode);
}

// chunks needed by the caller to size `part`:
int pdta_bn_chunks(long rows, int C) {
  int rpb, chunks;
  reduce_grid(rows, C, &rpb, &chunks);
  return chunks + 64;  // +64 rows of mid-reduce scratch at the tail
}

void pdta_bn_apply(const void* x, void* y, const void* res,
                   const float* weight, const float* bias, const float* mean,
                   const float* invstd, long rows, int C, int relu, int dt,
                   unsigned char* mask_out, hipStream_t s) {
  dim3 g = ew_grid2(rows, C);
  PDTA_DISPATCH_DTYPE(dt, T, {
    if (res) {
      if (relu)
        bn_apply_k<T, true, true><<<g, TPB, 0, s>>>(
            (const T*)x, (T*)y, (const T*)res, weight, bias, mean, invstd,
            rows, C / 8, mask_out);
      else
        bn_apply_k<T, false, true><<<g, TPB, 0, s>>>(
            (const T*)x, (T*)y, (const T*)res, weight, bias, mean, invstd,
            rows, C / 8);
    } else {
      if (relu)
        bn_apply_k<T, true, false><<<g, TPB, 0, s>>>(
            (const T*)x, (T*)y, nullptr, weight, bias, mean, invstd, rows,
            C / 8);
      else
        bn_apply_k<T, false, false><<<g, TPB, 0, s>>>(
            (const T*)x, (T*)y, nullptr, weight, bias, mean, invstd, rows,
            C / 8);
    }
  });
}

// mask: 0 none; 1 read relu_y; 2 recompute gamma*xhat+beta > 0 (needs
// invstd/weight/bias, skips the relu_y HBM pass)
void pdta_bn_bwd_reduce(const void* x, const void* dy, const void* relu_y,
                        const float* mean, const float* invstd,
                        const float* weight, const float* bias, int mask,
                        float* part, float* sum_dy,
                        float* sum_dy_xmu, long rows, int C, int dt,
                        const void* dy2, void* dres_out,
                        const unsigned char* mbits, hipStream_t s) {
  int rpb, chunks;
  dim3 g = reduce_grid(rows, C, &rpb, &chunks);
  PDTA_DISPATCH_DTYPE(dt, T, {
    if (mbits && dy2)
      // 1-bit join mask + two-dy: 1/16 the mask bytes of MASK 1
      bn_bwd_part_k<T, 3, true><<<g, TPB, 0, s>>>(
          (const T*)x, (const T*)dy, nullptr, mean, nullptr,
          nullptr, nullptr, part, rows, C, rpb, (const T*)dy2,
          (T*)dres_out, mbits);
    else if (mbits)
      bn_bwd_part_k<T, 3><<<g, TPB, 0, s>>>(
          (const T*)x, (const T*)dy, nullptr, mean, nullptr, nullptr,
          nullptr, part, rows, C, rpb, nullptr, nullptr, mbits);
    else if (mask == 1 && dy2)
      // two-dy: residual-fork gradient summed on load, masked combined
      // dy written out as dres
      bn_bwd_part_k<T, 1, true><<<g, TPB, 0, s>>>(
          (const T*)x, (const T*)dy, (const T*)relu_y, mean, nullptr,
          nullptr, nullptr, part, rows, C, rpb, (const T*)dy2,
          (T*)dres_out);
    else if (mask == 1)
      bn_bwd_part_k<T, 1><<<g, TPB, 0, s>>>(
          (const T*)x, (const T*)dy, (const T*)relu_y, mean, nullptr,
          nullptr, nullptr, part, rows, C, rpb);
    else if (mask == 2)
      bn_bwd_part_k<T, 2><<<g, TPB, 0, s>>>(
          (const T*)x, (const T*)dy, nullptr, mean, invstd, weight, bias,
          part, rows, C, rpb);
    else
      bn_bwd_part_k<T, 0><<<g, TPB, 0, s>>>(
          (const T*)x, (const T*)dy, nullptr, mean, nullptr, nullptr,
          nullptr, part, rows, C, rpb);
  });
  const float* fin = part;
  int fchunks = chunks;
  if (chunks > 64) {
    float* part2 = part + (long)chunks * 2 * C;
    bn_mid_k<<<dim3((2 * C + 255) / 256, 64), 256, 0, s>>>(part, part2,
                                                           chunks, 2 * C);
    fin = part2;
    fchunks = 64;
  }
  bn_bwd_final_k<<<(C + 63) / 64, 256, 0, s>>>(fin, fchunks, C, sum_dy,
                                               sum_dy_xmu);
}

void pdta_bn_bwd_elemt(const void* x, const void* dy, const void* relu_y,
                       void* dx, void* dres, const float* weight,
                       const float* bias, int mask,
                       const float* mean, const float* invstd,
                       const float* sum_dy, const float* sum_dy_xmu,
                       float count, long rows, int C, int dt,
                       const void* dy2, hipStream_t s) {
  dim3 g = ew_grid2(rows, C);
  PDTA_DISPATCH_DTYPE(dt, T, {
    if (mask == 1) {
      if (dres && dy2)
        bn_bwd_elemt_k<T, 1, true, true><<<g, TPB, 0, s>>>(
            (const T*)x, (const T*)dy, (const T*)relu_y, (T*)dx, (T*)dres,
            weight, nullptr, mean, invstd, sum_dy, sum_dy_xmu, 1.f / count,
            rows, C / 8, (const T*)dy2);
      else if (dres)
        bn_bwd_elemt_k<T, 1, true><<<g, TPB, 0, s>>>(
            (const T*)x, (const T*)dy, (const T*)relu_y, (T*)dx, (T*)dres,
            weight, nullptr, mean, invstd, sum_dy, sum_dy_xmu, 1.f / count,
            rows, C / 8);
      else if (dy2)
        bn_bwd_elemt_k<T, 1, false, true><<<g, TPB, 0, s>>>(
            (const T*)x, (const T*)dy, (const T*)relu_y, (T*)dx, nullptr,
            weight, nullptr, mean, invstd, sum_dy, sum_dy_xmu, 1.f / count,
            rows, C / 8, (const T*)dy2);
      else
        bn_bwd_elemt_k<T, 1, false><<<g, TPB, 0, s>>>(
            (const T*)x, (const T*)dy, (const T*)relu_y, (T*)dx, nullptr,
            weight, nullptr, mean, invstd, sum_dy, sum_dy_xmu, 1.f / count,
            rows, C / 8);
    } else if (mask == 2) {
      bn_bwd_elemt_k<T, 2, false><<<g, TPB, 0, s>>>(
          (const T*)x, (const T*)dy, nullptr, (T*)dx, nullptr, weight, bias,
          mean, invstd, sum_dy, sum_dy_xmu, 1.f / count, rows, C / 8);
    } else {
      bn_bwd_elemt_k<T, 0, false><<<g, TPB, 0, s>>>(
          (const T*)x, (const T*)dy, nullptr, (T*)dx, nullptr, weight,
          nullptr, mean, invstd, sum_dy, sum_dy_xmu, 1.f / count, rows,
          C / 8);
    }
  });
}


void pdta_syncbn_pack(const float* mean, const float* var, float n,
                      float* out, int C, hipStream_t s) {
  syncbn_pack_k<<<(C + 255) / 256, 256, 0, s>>>(mean, var, n, out, C);
}

void pdta_syncbn_combine(const float* g, int ws, int C, float eps,
                         float momentum, float* mean, float* invstd,
                         float* running_mean, float* running_var,
                         float* total_out, hipStream_t s) {
  syncbn_combine_k<<<(C + 255) / 256, 256, 0, s>>>(
      g, ws, C, eps, momentum, mean, invstd, running_mean, running_var,
      total_out);
}

void pdta_syncbn_pack2(const float* a, const float* b, float n, float* out,
                       int C, hipStream_t s) {
  syncbn_pack2_k<<<(C + 255) / 256, 256, 0, s>>>(a, b, n, out, C);
}

void pdta_syncbn_div(float* p, int C, hipStream_t s) {
  syncbn_div_k<<<(2 * C + 255) / 256, 256, 0, s>>>(p, C);
}

}  // extern "C"
