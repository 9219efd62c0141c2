// Transformer pointwise kernels for the ViT path (SURVEY.md §2d softmax/
// LayerNorm/GELU row; BASELINE config #5): LayerNorm fwd/bwd over rows of
// d (wave-per-row, d % 64 == 0), exact-erf GELU fwd/bwd, masked+scaled
// row softmax fwd/bwd (attention probabilities; cols >= nvalid are -inf
// so sequence padding to a GEMM-friendly length stays exact).
#include "common.h"

namespace {

constexpr int TPB = 256;  // 4 waves -> 4 rows per block

// ---- LayerNorm ----------------------------------------------------------
template <typename T>
__global__ void ln_fwd_k(const T* __restrict__ x, T* __restrict__ y,
                         const float* __restrict__ w,
                         const float* __restrict__ b,
                         float* __restrict__ mean_out,
                         float* __restrict__ invstd_out, long rows, int d,
                         float eps) {
  long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  int lane = threadIdx.x & 63;
  const T* xr = x + row * d;
  float s = 0.f, ss = 0.f;
  for (int c = lane * 8; c < d; c += 64 * 8) {
    Vec8<T> v = load8(xr + c);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = to_f32(v.v[j]);
      s += f;
      ss += f * f;
    }
  }
  s = wave_reduce_sum(s);
  ss = wave_reduce_sum(ss);
  float mu = s / d;
  float var = fmaxf(ss / d - mu * mu, 0.f);
  float is = rsqrtf(var + eps);
  if (lane == 0) {
    mean_out[row] = mu;
    invstd_out[row] = is;
  }
  T* yr = y + row * d;
  for (int c = lane * 8; c < d; c += 64 * 8) {
    Vec8<T> v = load8(xr + c);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      v.v[j] = from_f32<T>((to_f32(v.v[j]) - mu) * is * w[c + j] + b[c + j]);
    store8(yr + c, v);
  }
}

// dx = is * w .* (dy - mean(dy.*w) ... ) per row; dw/db partials are
// accumulated in REGISTERS per thread (a per-element LDS-atomic version
// was 261 us/call), combined across the block's 4 waves through LDS once
// per chunk, written to part [chunk][2d]; ln_bwd_final reduces chunks
// with 64-col x 4-slice blocks (a d-threads-only finalize was one wave
// chip-wide and chunk-serial).
template <typename T>
__global__ void ln_bwd_k(const T* __restrict__ x, const T* __restrict__ dy,
                         T* __restrict__ dx, const float* __restrict__ w,
                         const float* __restrict__ mean,
                         const float* __restrict__ invstd,
                         float* __restrict__ part, long rows, int d,
                         long rows_per_chunk) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* red = reinterpret_cast<float*>(smem);  // [4 waves][2d]

  long r0 = (long)blockIdx.x * rows_per_chunk;
  long r1 = min(rows, r0 + rows_per_chunk);
  int lane = threadIdx.x & 63;
  int wv = threadIdx.x >> 6;
  constexpr int MAXG = 8;           // supports d up to 4096
  int ngroups = (d + 511) / 512;
  float adw[MAXG][8], adb[MAXG][8];
#pragma unroll
  for (int g = 0; g < MAXG; ++g)
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      adw[g][j] = 0.f;
      adb[g][j] = 0.f;
    }

  for (long row = r0 + wv; row < r1; row += 4) {
    const T* xr = x + row * d;
    const T* gr = dy + row * d;
    float mu = mean[row], is = invstd[row];
    float s1 = 0.f, s2 = 0.f;  // sum(dy*w), sum(dy*w*xhat)
    for (int c = lane * 8; c < d; c += 512) {
      Vec8<T> g = load8(gr + c);
      Vec8<T> v = load8(xr + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float gw = to_f32(g.v[j]) * w[c + j];
        float xh = (to_f32(v.v[j]) - mu) * is;
        s1 += gw;
        s2 += gw * xh;
      }
    }
    s1 = wave_reduce_sum(s1) / d;
    s2 = wave_reduce_sum(s2) / d;
    T* dr = dx + row * d;
    int g = 0;
    for (int c = lane * 8; c < d; c += 512, ++g) {
      Vec8<T> gv = load8(gr + c);
      Vec8<T> v = load8(xr + c);
      Vec8<T> o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float gg = to_f32(gv.v[j]);
        float xh = (to_f32(v.v[j]) - mu) * is;
        o.v[j] = from_f32<T>(is * (gg * w[c + j] - s1 - xh * s2));
        adw[g][j] += gg * xh;
        adb[g][j] += gg;
      }
      store8(dr + c, o);
    }
  }
  // combine the 4 waves through LDS (no atomics), one pass per chunk
  float* my = red + wv * 2 * d;
  {
    int g = 0;
    for (int c = lane * 8; c < d; c += 512, ++g)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        my[c + j] = adw[g][j];
        my[d + c + j] = adb[g][j];
      }
  }
  __syncthreads();
  float* out = part + (long)blockIdx.x * 2 * d;
  for (int i = threadIdx.x; i < 2 * d; i += blockDim.x)
    out[i] = red[i] + red[2 * d + i] + red[4 * d + i] + red[6 * d + i];
}

__global__ void ln_bwd_final_k(const float* __restrict__ part, int chunks,
                               int d, float* __restrict__ dw,
                               float* __restrict__ db) {
  __shared__ float red[2][256];
  int c = blockIdx.x * 64 + (int)(threadIdx.x % 64);
  int slice = threadIdx.x / 64;
  float sw = 0.f, sb = 0.f;
  if (c < d) {
    for (int k = slice; k < chunks; k += 4) {
      sw += part[(long)k * 2 * d + c];
      sb += part[(long)k * 2 * d + d + c];
    }
  }
  red[0][threadIdx.x] = sw;
  red[1][threadIdx.x] = sb;
  __syncthreads();
  if (slice == 0 && c < d) {
#pragma unroll
    for (int i = 1; i < 4; ++i) {
      sw += red[0][threadIdx.x + i * 64];
      sb += red[1][threadIdx.x + i * 64];
    }
    dw[c] = sw;
    db[c] = sb;
  }
}

// ---- GELU (exact erf form, matching torch GELU default) -----------------
template <typename T>
__global__ void gelu_fwd_k(const T* __restrict__ x, T* __restrict__ y,
                           long n8) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n8; i += stride) {
    Vec8<T> v = load8(x + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = to_f32(v.v[j]);
      v.v[j] = from_f32<T>(0.5f * f * (1.f + erff(f * 0.70710678f)));
    }
    store8(y + i * 8, v);
  }
}

template <typename T>
__global__ void gelu_bwd_k(const T* __restrict__ x, const T* __restrict__ dy,
                           T* __restrict__ dx, long n8) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n8; i += stride) {
    Vec8<T> v = load8(x + i * 8);
    Vec8<T> g = load8(dy + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = to_f32(v.v[j]);
      float cdf = 0.5f * (1.f + erff(f * 0.70710678f));
      float pdf = 0.39894228f * __expf(-0.5f * f * f);
      g.v[j] = from_f32<T>(to_f32(g.v[j]) * (cdf + f * pdf));
    }
    store8(dx + i * 8, g);
  }
}

// ---- masked, scaled row softmax (attention probabilities) ---------------
// in: fp32 scores [rows][n]; out: dtype probs. cols >= nvalid get 0.
__global__ void softmax_fwd_f32_k(const float* __restrict__ s,
                                  void* __restrict__ p_out, long rows,
                                  int n, int nvalid, float scale, int dt) {
  // vectorized: lane handles 4 f32 cols per step (n % 256 == 0 here)
  long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  int lane = threadIdx.x & 63;
  const float4* sr = reinterpret_cast<const float4*>(s + row * n);
  int n4 = n >> 2;
  float mx = -3.4e38f;
  for (int c = lane; c < n4; c += 64) {
    float4 v = sr[c];
    int c0 = c * 4;
    if (c0 + 0 < nvalid) mx = fmaxf(mx, v.x);
    if (c0 + 1 < nvalid) mx = fmaxf(mx, v.y);
    if (c0 + 2 < nvalid) mx = fmaxf(mx, v.z);
    if (c0 + 3 < nvalid) mx = fmaxf(mx, v.w);
  }
  mx = wave_reduce_max(mx) * scale;
  float sum = 0.f;
  for (int c = lane; c < n4; c += 64) {
    float4 v = sr[c];
    int c0 = c * 4;
    sum += (c0 + 0 < nvalid) ? __expf(v.x * scale - mx) : 0.f;
    sum += (c0 + 1 < nvalid) ? __expf(v.y * scale - mx) : 0.f;
    sum += (c0 + 2 < nvalid) ? __expf(v.z * scale - mx) : 0.f;
    sum += (c0 + 3 < nvalid) ? __expf(v.w * scale - mx) : 0.f;
  }
  sum = wave_reduce_sum(sum);
  float inv = 1.f / sum;
  if (dt == PDTA_DT_BF16) {
    __hip_bfloat16* pr = (__hip_bfloat16*)p_out + row * n;
    for (int c = lane; c < n4; c += 64) {
      float4 v = sr[c];
      int c0 = c * 4;
      __hip_bfloat16 o[4];
      o[0] = from_f32<__hip_bfloat16>(
          c0 + 0 < nvalid ? __expf(v.x * scale - mx) * inv : 0.f);
      o[1] = from_f32<__hip_bfloat16>(
          c0 + 1 < nvalid ? __expf(v.y * scale - mx) * inv : 0.f);
      o[2] = from_f32<__hip_bfloat16>(
          c0 + 2 < nvalid ? __expf(v.z * scale - mx) * inv : 0.f);
      o[3] = from_f32<__hip_bfloat16>(
          c0 + 3 < nvalid ? __expf(v.w * scale - mx) * inv : 0.f);
      *reinterpret_cast<uint2*>(pr + c0) = *reinterpret_cast<uint2*>(o);
    }
  } else {
    __half* pr = (__half*)p_out + row * n;
    for (int c = lane; c < n4; c += 64) {
      float4 v = sr[c];
      int c0 = c * 4;
      __half o[4];
      o[0] = from_f32<__half>(
          c0 + 0 < nvalid ? __expf(v.x * scale - mx) * inv : 0.f);
      o[1] = from_f32<__half>(
          c0 + 1 < nvalid ? __expf(v.y * scale - mx) * inv : 0.f);
      o[2] = from_f32<__half>(
          c0 + 2 < nvalid ? __expf(v.z * scale - mx) * inv : 0.f);
      o[3] = from_f32<__half>(
          c0 + 3 < nvalid ? __expf(v.w * scale - mx) * inv : 0.f);
      *reinterpret_cast<uint2*>(pr + c0) = *reinterpret_cast<uint2*>(o);
    }
  }
}

// dscores = scale * p .* (dp - sum(p .* dp)), emitted in compute dtype
template <typename T>
__global__ void softmax_bwd_k(const T* __restrict__ p, const T* __restrict__ dp,
                              T* __restrict__ ds, long rows, int n,
                              float scale) {
  long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  int lane = threadIdx.x & 63;
  const T* pr = p + row * n;
  const T* gr = dp + row * n;
  int n8 = n >> 3;
  float dot = 0.f;
  for (int c = lane; c < n8; c += 64) {
    Vec8<T> pv = load8(pr + c * 8);
    Vec8<T> gv = load8(gr + c * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) dot += to_f32(pv.v[j]) * to_f32(gv.v[j]);
  }
  dot = wave_reduce_sum(dot);
  T* dr = ds + row * n;
  for (int c = lane; c < n8; c += 64) {
    Vec8<T> pv = load8(pr + c * 8);
    Vec8<T> gv = load8(gr + c * 8);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o.v[j] = from_f32<T>(scale * to_f32(pv.v[j]) *
                           (to_f32(gv.v[j]) - dot));
    store8(dr + c * 8, o);
  }
}

}  // namespace

extern "C" {

void pdta_ln_fwd(const void* x, void* y, const float* w, const float* b,
                 float* mean, float* invstd, long rows, int d, float eps,
                 int dt, hipStream_t s) {
  long blocks = (rows + 3) / 4;
  PDTA_DISPATCH_DTYPE16(dt, T, {
    ln_fwd_k<T><<<(unsigned)blocks, TPB, 0, s>>>(
        (const T*)x, (T*)y, w, b, mean, invstd, rows, d, eps);
  });
}

int pdta_ln_chunks(long rows) {
  long chunks = 1024;
  long rpc = (rows + chunks - 1) / chunks;
  if (rpc < 4) rpc = 4;
  return (int)((rows + rpc - 1) / rpc);
}

void pdta_ln_bwd(const void* x, const void* dy, void* dx, const float* w,
                 const float* mean, const float* invstd, float* part,
                 float* dw, float* db, long rows, int d, int dt,
                 hipStream_t s) {
  int chunks = pdta_ln_chunks(rows);
  long rpc = (rows + chunks - 1) / chunks;
  int lds = 4 * 2 * d * sizeof(float);
  PDTA_DISPATCH_DTYPE16(dt, T, {
    ln_bwd_k<T><<<chunks, TPB, lds, s>>>((const T*)x, (const T*)dy, (T*)dx,
                                         w, mean, invstd, part, rows, d,
                                         rpc);
  });
  ln_bwd_final_k<<<(d + 63) / 64, 256, 0, s>>>(part, chunks, d, dw, db);
}

void pdta_gelu_fwd(const void* x, void* y, long n, int dt, hipStream_t s) {
  long n8 = n / 8;
  PDTA_DISPATCH_DTYPE16(dt, T, {
    gelu_fwd_k<T><<<ew_grid(n8, TPB, 1), TPB, 0, s>>>((const T*)x, (T*)y,
                                                      n8);
  });
}

void pdta_gelu_bwd(const void* x, const void* dy, void* dx, long n, int dt,
                   hipStream_t s) {
  long n8 = n / 8;
  PDTA_DISPATCH_DTYPE16(dt, T, {
    gelu_bwd_k<T><<<ew_grid(n8, TPB, 1), TPB, 0, s>>>(
        (const T*)x, (const T*)dy, (T*)dx, n8);
  });
}

void pdta_softmax_fwd(const float* scores, void* probs, long rows, int n,
                      int nvalid, float scale, int dt, hipStream_t s) {
  long blocks = (rows + 3) / 4;
  softmax_fwd_f32_k<<<(unsigned)blocks, TPB, 0, s>>>(scores, probs, rows, n,
                                                     nvalid, scale, dt);
}

void pdta_softmax_bwd(const void* probs, const void* dprobs, void* dscores,
                      long rows, int n, float scale, int dt, hipStream_t s) {
  long blocks = (rows + 3) / 4;
  PDTA_DISPATCH_DTYPE16(dt, T, {
    softmax_bwd_k<T><<<(unsigned)blocks, TPB, 0, s>>>(
        (const T*)probs, (const T*)dprobs, (T*)dscores, rows, n, scale);
  });
}

}  // extern "C"
