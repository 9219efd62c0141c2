// BatchNorm kernels over [N,H,W,C] (C innermost, C % 64 == 0).
// Rebuilds torch's batch_norm_stats / batch_norm_elemt /
// batch_norm_backward_reduce / batch_norm_backward_elemt set
// (SURVEY.md §2d BatchNorm row) as NHWC-native CDNA4 kernels:
//   - stats & bwd_reduce: each thread owns ONE channel column and strides
//     rows (coalesced: consecutive lanes read consecutive channels),
//     fp32 accumulation, per-block atomicAdd into fp32 [C] buffers;
//   - apply & bwd_elemt: each thread owns EIGHT channels (16-B bf16
//     vectors), per-channel affine constants held in registers across the
//     row loop — pure HBM-rate streaming with optional fused ReLU.
// The same kernels serve SyncBatchNorm: stats deliver (sum, sumsq) partials
// that the python layer all-gathers/combines.
#include "common.h"

namespace {

constexpr int TPB = 256;

// -- per-channel reduction: sum and sumsq --------------------------------
template <typename T>
__global__ void bn_stats_k(const T* __restrict__ x, float* __restrict__ sum,
                           float* __restrict__ sumsq, long rows, int C,
                           int rows_per_block) {
  int cpb = min(C, TPB);            // channels covered per block
  int rpar = TPB / cpb;             // rows processed in parallel
  int c = blockIdx.x * cpb + (int)(threadIdx.x % cpb);
  long r0 = (long)blockIdx.y * rows_per_block + threadIdx.x / cpb;
  long r_end = min(rows, (long)(blockIdx.y + 1) * rows_per_block);
  float s = 0.f, ss = 0.f;
  for (long r = r0; r < r_end; r += rpar) {
    float v = to_f32(x[r * C + c]);
    s += v;
    ss += v * v;
  }
  atomicAdd(&sum[c], s);
  atomicAdd(&sumsq[c], ss);
}

__global__ void bn_finalize_k(const float* __restrict__ sum,
                              const float* __restrict__ sumsq,
                              float* __restrict__ mean,
                              float* __restrict__ var, float inv_n, int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c < C) {
    float m = sum[c] * inv_n;
    mean[c] = m;
    var[c] = fmaxf(sumsq[c] * inv_n - m * m, 0.f);
  }
}

// -- y = (x - mean) * invstd * w + b, optional ReLU ----------------------
// thread mapping for the streaming kernels: cpg = min(C/8, TPB) threads
// cover the channel axis (8 channels each), the rest of the block plus
// gridDim.y stride the row axis; per-thread channel constants live in
// registers across the whole row loop.
template <typename T, bool RELU>
__global__ void bn_apply_k(const T* __restrict__ x, T* __restrict__ y,
                           const float* __restrict__ weight,
                           const float* __restrict__ bias,
                           const float* __restrict__ mean,
                           const float* __restrict__ invstd, long rows,
                           int c8) {  // c8 = C/8
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
  for (long r = (long)blockIdx.y * rpar + threadIdx.x / cpg; r < rows;
       r += rstride) {
    Vec8<T> v = load8(x + r * C + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float o = to_f32(v.v[j]) * scale[j] + shift[j];
      if (RELU) o = fmaxf(o, 0.f);
      v.v[j] = from_f32<T>(o);
    }
    store8(y + r * C + c0, v);
  }
}

// -- backward reduce: sum_dy, sum_dy_xmu ---------------------------------
template <typename T>
__global__ void bn_bwd_reduce_k(const T* __restrict__ x,
                                const T* __restrict__ dy,
                                const float* __restrict__ mean,
                                float* __restrict__ sum_dy,
                                float* __restrict__ sum_dy_xmu, long rows,
                                int C, int rows_per_block) {
  int cpb = min(C, TPB);
  int rpar = TPB / cpb;
  int c = blockIdx.x * cpb + (int)(threadIdx.x % cpb);
  long r0 = (long)blockIdx.y * rows_per_block + threadIdx.x / cpb;
  long r_end = min(rows, (long)(blockIdx.y + 1) * rows_per_block);
  float m = mean[c];
  float s = 0.f, sx = 0.f;
  for (long r = r0; r < r_end; r += rpar) {
    float g = to_f32(dy[r * C + c]);
    float v = to_f32(x[r * C + c]);
    s += g;
    sx += g * (v - m);
  }
  atomicAdd(&sum_dy[c], s);
  atomicAdd(&sum_dy_xmu[c], sx);
}

// -- backward elementwise:
//    dx = w*invstd * (dy - sum_dy/n - (x-mean)*invstd^2*sum_dy_xmu/n)
template <typename T>
__global__ void bn_bwd_elemt_k(const T* __restrict__ x,
                               const T* __restrict__ dy, T* __restrict__ dx,
                               const float* __restrict__ weight,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               const float* __restrict__ sum_dy,
                               const float* __restrict__ sum_dy_xmu,
                               float inv_n, long rows, int c8) {
  int cpg = min(c8, TPB);
  int rpar = TPB / cpg;
  int cg = blockIdx.x * cpg + (int)(threadIdx.x % cpg);
  if (cg >= c8) return;
  int c0 = cg * 8;
  float k1[8], k2[8], k3[8], mu[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float is = invstd[c0 + j];
    k1[j] = weight[c0 + j] * is;
    k2[j] = sum_dy[c0 + j] * inv_n;
    k3[j] = is * is * sum_dy_xmu[c0 + j] * inv_n;
    mu[j] = mean[c0 + j];
  }
  const int C = c8 * 8;
  long rstride = (long)gridDim.y * rpar;
  for (long r = (long)blockIdx.y * rpar + threadIdx.x / cpg; r < rows;
       r += rstride) {
    Vec8<T> g = load8(dy + r * C + c0);
    Vec8<T> v = load8(x + r * C + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float o = k1[j] * (to_f32(g.v[j]) - k2[j] -
                         (to_f32(v.v[j]) - mu[j]) * k3[j]);
      g.v[j] = from_f32<T>(o);
    }
    store8(dx + r * C + c0, g);
  }
}

static inline dim3 reduce_grid(long rows, int C, int* rows_per_block) {
  int cpb = C < TPB ? C : TPB;
  int cblocks = (C + cpb - 1) / cpb;
  // target ~2048 blocks total for full-chip residency
  long want = 2048 / cblocks;
  if (want < 1) want = 1;
  long rpb = (rows + want - 1) / want;
  int rpar = TPB / cpb;
  if (rpb < rpar) rpb = rpar;
  *rows_per_block = (int)rpb;
  long rblocks = (rows + rpb - 1) / rpb;
  return dim3(cblocks, (unsigned)rblocks);
}

static inline dim3 ew_grid2(long rows, int C) {
  int c8 = C / 8;
  int cpg = c8 < TPB ? c8 : TPB;
  int cgb = (c8 + cpg - 1) / cpg;  // blocks along channel axis
  int rpar = TPB / cpg;
  long rblocks = (rows + rpar - 1) / rpar;
  long want = 2048 / cgb;  // ~2048 blocks total
  if (want < 1) want = 1;
  if (rblocks > want) rblocks = want;
  return dim3(cgb, (unsigned)rblocks);
}

}  // namespace

extern "C" {

// sum/sumsq/mean/var are fp32 [C]; sum & sumsq must be ZEROED by caller.
void pdta_bn_stats(const void* x, float* sum, float* sumsq, float* mean,
                   float* var, long rows, int C, int dt, hipStream_t s) {
  int rpb;
  dim3 g = reduce_grid(rows, C, &rpb);
  PDTA_DISPATCH_DTYPE(dt, T, {
    bn_stats_k<T><<<g, TPB, 0, s>>>((const T*)x, sum, sumsq, rows, C, rpb);
  });
  bn_finalize_k<<<(C + 255) / 256, 256, 0, s>>>(sum, sumsq, mean, var,
                                                1.f / (float)rows, C);
}

void pdta_bn_apply(const void* x, void* y, const float* weight,
                   const float* bias, const float* mean, const float* invstd,
                   long rows, int C, int relu, int dt, hipStream_t s) {
  dim3 g = ew_grid2(rows, C);
  PDTA_DISPATCH_DTYPE(dt, T, {
    if (relu)
      bn_apply_k<T, true><<<g, TPB, 0, s>>>((const T*)x, (T*)y, weight, bias,
                                            mean, invstd, rows, C / 8);
    else
      bn_apply_k<T, false><<<g, TPB, 0, s>>>((const T*)x, (T*)y, weight,
                                             bias, mean, invstd, rows, C / 8);
  });
}

// sum_dy / sum_dy_xmu fp32 [C], ZEROED by caller.
void pdta_bn_bwd_reduce(const void* x, const void* dy, const float* mean,
                        float* sum_dy, float* sum_dy_xmu, long rows, int C,
                        int dt, hipStream_t s) {
  int rpb;
  dim3 g = reduce_grid(rows, C, &rpb);
  PDTA_DISPATCH_DTYPE(dt, T, {
    bn_bwd_reduce_k<T><<<g, TPB, 0, s>>>((const T*)x, (const T*)dy, mean,
                                         sum_dy, sum_dy_xmu, rows, C, rpb);
  });
}

void pdta_bn_bwd_elemt(const void* x, const void* dy, void* dx,
                       const float* weight, const float* mean,
                       const float* invstd, const float* sum_dy,
                       const float* sum_dy_xmu, float count, long rows,
                       int C, int dt, hipStream_t s) {
  dim3 g = ew_grid2(rows, C);
  PDTA_DISPATCH_DTYPE(dt, T, {
    bn_bwd_elemt_k<T><<<g, TPB, 0, s>>>(
        (const T*)x, (const T*)dy, (T*)dx, weight, mean, invstd, sum_dy,
        sum_dy_xmu, 1.f / count, rows, C / 8);
  });
}

}  // extern "C"
