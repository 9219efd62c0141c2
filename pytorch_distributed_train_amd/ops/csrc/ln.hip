// Transformer pointwise kernels for the ViT path (SURVEY.md §2d softmax/
// LayerNorm/GELU row; BASELINE config #5): LayerNorm fwd/bwd over rows of
// d (wave-per-row, d % 64 == 0), exact-erf GELU fwd/bwd, masked+scaled
// row softmax fwd/bwd (attention probabilities; cols >= nvalid are -inf
// so sequence padding to a GEMM-friendly length stays exact).
#include "common.h"

namespace {

constexpr int TPB = 256;  // 4 waves -> 4 rows per block

// ---- LayerNorm ----------------------------------------------------------
// NG = number of 512-column groups covering d (register arrays are sized
// exactly: a fixed max-size version cost 128 VGPRs of dead partials and
// single-wave occupancy — 6x off the HBM traffic floor in rocprof).
// The row is loaded ONCE into registers and reused for the apply pass.
template <typename T, int NG, bool RES = false>
__global__ void ln_fwd_k(const T* __restrict__ x, T* __restrict__ y,
                         const float* __restrict__ w,
                         const float* __restrict__ b,
                         float* __restrict__ mean_out,
                         float* __restrict__ invstd_out, long rows, int d,
                         float eps, const T* __restrict__ res = nullptr,
                         T* __restrict__ sum_out = nullptr) {
  // RES: fused residual join ln(x + res), writing the (rounded) sum out
  // too — replaces the ViT block's separate add kernel AND its autograd
  // fork (the sum has exactly one consumer downstream)
  long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  int lane = threadIdx.x & 63;
  const T* xr = x + row * d;
  Vec8<T> xv[NG];
  float s = 0.f, ss = 0.f;
#pragma unroll
  for (int g = 0; g < NG; ++g) {
    int c = lane * 8 + g * 512;
    if (c < d) {
      xv[g] = load8(xr + c);
      if (RES) {
        Vec8<T> rv = load8(res + row * d + c);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          xv[g].v[j] = from_f32<T>(to_f32(xv[g].v[j]) + to_f32(rv.v[j]));
        store8(sum_out + row * d + c, xv[g]);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = to_f32(xv[g].v[j]);
        s += f;
        ss += f * f;
      }
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
#pragma unroll
  for (int g = 0; g < NG; ++g) {
    int c = lane * 8 + g * 512;
    if (c < d) {
      float4 w0 = *reinterpret_cast<const float4*>(w + c);
      float4 w1 = *reinterpret_cast<const float4*>(w + c + 4);
      float4 b0 = *reinterpret_cast<const float4*>(b + c);
      float4 b1 = *reinterpret_cast<const float4*>(b + c + 4);
      const float* wp = reinterpret_cast<const float*>(&w0);
      const float* bp = reinterpret_cast<const float*>(&b0);
      Vec8<T> o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float wj = j < 4 ? wp[j] : reinterpret_cast<const float*>(&w1)[j - 4];
        float bj = j < 4 ? bp[j] : reinterpret_cast<const float*>(&b1)[j - 4];
        o.v[j] = from_f32<T>((to_f32(xv[g].v[j]) - mu) * is * wj + bj);
      }
      store8(yr + c, o);
    }
  }
}

// dx = is * w .* (dy - mean(dy.*w) ... ) per row; dw/db partials are
// accumulated in REGISTERS per thread (a per-element LDS-atomic version
// was 261 us/call), combined across the block's 4 waves through LDS once
// per chunk, written to part [chunk][2d]; ln_bwd_final reduces chunks
// with 64-col x 4-slice blocks (a d-threads-only finalize was one wave
// chip-wide and chunk-serial).
// __launch_bounds__(TPB): without it the compiler budgets registers for
// a hypothetical 1024-thread block (4 waves/SIMD floor -> 128-VGPR cap)
// and SPILLS 112-124 B/lane of the register-cached row (NG=2, d=768) to
// scratch EVERY row iteration — measured 104 us/call vs the ~45 us this
// kernel's 4-5 streams cost at ln_fwd's 4 TB/s. Declaring the real
// 256-thread shape lifts the cap and the spills vanish.
template <typename T, int NG, bool DXADD = false>
__global__ __launch_bounds__(256) void ln_bwd_k(
    const T* __restrict__ x, const T* __restrict__ dy,
                         T* __restrict__ dx, const float* __restrict__ w,
                         const float* __restrict__ mean,
                         const float* __restrict__ invstd,
                         float* __restrict__ part, long rows, int d,
                         long rows_per_chunk,
                         const T* __restrict__ dxadd = nullptr) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* red = reinterpret_cast<float*>(smem);  // [4 waves][2d]

  long r0 = (long)blockIdx.x * rows_per_chunk;
  long r1 = min(rows, r0 + rows_per_chunk);
  int lane = threadIdx.x & 63;
  int wv = threadIdx.x >> 6;
  float adw[NG][8], adb[NG][8];
  float wreg[NG][8];  // w is row-invariant: load once per kernel
#pragma unroll
  for (int g = 0; g < NG; ++g) {
    int c = lane * 8 + g * 512;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      adw[g][j] = 0.f;
      adb[g][j] = 0.f;
      wreg[g][j] = 0.f;
    }
    if (c < d) {
      float4 w0 = *reinterpret_cast<const float4*>(w + c);
      float4 w1 = *reinterpret_cast<const float4*>(w + c + 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        wreg[g][j] = reinterpret_cast<const float*>(&w0)[j];
        wreg[g][j + 4] = reinterpret_cast<const float*>(&w1)[j];
      }
    }
  }

  for (long row = r0 + wv; row < r1; row += 4) {
    const T* xr = x + row * d;
    const T* gr = dy + row * d;
    float mu = mean[row], is = invstd[row];
    Vec8<T> gv[NG], xv[NG];  // row cached in registers: one HBM read
    float s1 = 0.f, s2 = 0.f;  // sum(dy*w), sum(dy*w*xhat)
#pragma unroll
    for (int g = 0; g < NG; ++g) {
      int c = lane * 8 + g * 512;
      if (c < d) {
        gv[g] = load8(gr + c);
        xv[g] = load8(xr + c);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float gw = to_f32(gv[g].v[j]) * wreg[g][j];
          float xh = (to_f32(xv[g].v[j]) - mu) * is;
          s1 += gw;
          s2 += gw * xh;
        }
      }
    }
    s1 = wave_reduce_sum(s1) / d;
    s2 = wave_reduce_sum(s2) / d;
    T* dr = dx + row * d;
#pragma unroll
    for (int g = 0; g < NG; ++g) {
      int c = lane * 8 + g * 512;
      if (c < d) {
        Vec8<T> o, av;
        if (DXADD) av = load8(dxadd + row * d + c);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float gg = to_f32(gv[g].v[j]);
          float xh = (to_f32(xv[g].v[j]) - mu) * is;
          float oo = is * (gg * wreg[g][j] - s1 - xh * s2);
          // DXADD: the downstream pass-through gradient of the fused
          // residual sum, folded into the dx store (kills the autograd
          // accumulation add at the fork)
          if (DXADD) oo += to_f32(av.v[j]);
          o.v[j] = from_f32<T>(oo);
          adw[g][j] += gg * xh;
          adb[g][j] += gg;
        }
        store8(dr + c, o);
      }
    }
  }
  // combine the 4 waves through LDS (no atomics), one pass per chunk
  float* my = red + wv * 2 * d;
#pragma unroll
  for (int g = 0; g < NG; ++g) {
    int c = lane * 8 + g * 512;
    if (c < d)
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

// Sums chunk rows s, s+nslab, s+2*nslab, ... of part into out[s][2d]
// (first level of the two-level finalize: with ~1000 chunks a single
// d/64-block finalize was 12 blocks chip-wide and latency-serial, 77 us).
__global__ void ln_bwd_psum_k(const float* __restrict__ part, int chunks,
                              int d, int nslab, float* __restrict__ out) {
  __shared__ float red[2][256];
  int c = blockIdx.x * 64 + (int)(threadIdx.x % 64);
  int slice = threadIdx.x / 64;
  int s = blockIdx.y;
  float sw = 0.f, sb = 0.f;
  if (c < d) {
    for (int k = s + slice * nslab; k < chunks; k += 4 * nslab) {
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
    out[(long)s * 2 * d + c] = sw;
    out[(long)s * 2 * d + d + c] = sb;
  }
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

// ---- GELU (erf form, matching torch GELU default) -----------------------
// erf via the Abramowitz-Stegun 7.1.26 rational approximation
// (|error| < 1.5e-7 — three orders below bf16 resolution): ~10 FMA + one
// __expf vs ocml erff's long polynomial (gelu was VALU-issue-bound).
DEV_INLINE float erf_fast(float x) {
  float ax = fabsf(x);
  float t = 1.f / fmaf(0.3275911f, ax, 1.f);
  float poly = t * fmaf(
      t, fmaf(t, fmaf(t, fmaf(t, 1.061405429f, -1.453152027f),
                      1.421413741f), -0.284496736f), 0.254829592f);
  float e = 1.f - poly * __expf(-ax * ax);
  return copysignf(e, x);
}

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
      v.v[j] = from_f32<T>(0.5f * f * (1.f + erf_fast(f * 0.70710678f)));
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
      float cdf = 0.5f * (1.f + erf_fast(f * 0.70710678f));
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
  if (n4 <= 64) {
    // row-in-registers fast path (n <= 256 — every ViT attention shape):
    // ONE read of the scores instead of three (the generic path was
    // measured exactly at its 3-read traffic floor)
    bool act = lane < n4;
    int c0 = lane * 4;
    float4 v = act ? sr[lane] : float4{0.f, 0.f, 0.f, 0.f};
    float e[4];
    float mx = -3.4e38f;
    if (act) {
      if (c0 + 0 < nvalid) mx = fmaxf(mx, v.x);
      if (c0 + 1 < nvalid) mx = fmaxf(mx, v.y);
      if (c0 + 2 < nvalid) mx = fmaxf(mx, v.z);
      if (c0 + 3 < nvalid) mx = fmaxf(mx, v.w);
    }
    mx = wave_reduce_max(mx) * scale;
    e[0] = (act && c0 + 0 < nvalid) ? __expf(v.x * scale - mx) : 0.f;
    e[1] = (act && c0 + 1 < nvalid) ? __expf(v.y * scale - mx) : 0.f;
    e[2] = (act && c0 + 2 < nvalid) ? __expf(v.z * scale - mx) : 0.f;
    e[3] = (act && c0 + 3 < nvalid) ? __expf(v.w * scale - mx) : 0.f;
    float sum = wave_reduce_sum(e[0] + e[1] + e[2] + e[3]);
    float inv = 1.f / sum;
    if (act) {
      if (dt == PDTA_DT_BF16) {
        __hip_bfloat16 o[4];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          o[j] = from_f32<__hip_bfloat16>(e[j] * inv);
        *reinterpret_cast<uint2*>((__hip_bfloat16*)p_out + row * n + c0) =
            *reinterpret_cast<uint2*>(o);
      } else {
        __half o[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) o[j] = from_f32<__half>(e[j] * inv);
        *reinterpret_cast<uint2*>((__half*)p_out + row * n + c0) =
            *reinterpret_cast<uint2*>(o);
      }
    }
    return;
  }
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
  if (n8 <= 64) {  // row cached in registers: one read of p and dp
    bool act = lane < n8;
    Vec8<T> pv, gv;
    float dot = 0.f;
    if (act) {
      pv = load8(pr + lane * 8);
      gv = load8(gr + lane * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) dot += to_f32(pv.v[j]) * to_f32(gv.v[j]);
    }
    dot = wave_reduce_sum(dot);
    if (act) {
      Vec8<T> o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o.v[j] = from_f32<T>(scale * to_f32(pv.v[j]) *
                             (to_f32(gv.v[j]) - dot));
      store8(ds + row * n + lane * 8, o);
    }
    return;
  }
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

// register-array groups covering d (columns handled per thread = NG*8)
#define PDTA_LN_DISPATCH_NG(d, NG, ...)        \
  if ((d) <= 512) {                            \
    constexpr int NG = 1;                      \
    __VA_ARGS__                                \
  } else if ((d) <= 1024) {                    \
    constexpr int NG = 2;                      \
    __VA_ARGS__                                \
  } else if ((d) <= 2048) {                    \
    constexpr int NG = 4;                      \
    __VA_ARGS__                                \
  } else {                                     \
    constexpr int NG = 8;                      \
    __VA_ARGS__                                \
  }

void pdta_ln_fwd(const void* x, void* y, const float* w, const float* b,
                 float* mean, float* invstd, long rows, int d, float eps,
                 int dt, const void* res, void* sum_out, hipStream_t s) {
  long blocks = (rows + 3) / 4;
  PDTA_DISPATCH_DTYPE16(dt, T, {
    PDTA_LN_DISPATCH_NG(d, NG, {
      if (res)
        ln_fwd_k<T, NG, true><<<(unsigned)blocks, TPB, 0, s>>>(
            (const T*)x, (T*)y, w, b, mean, invstd, rows, d, eps,
            (const T*)res, (T*)sum_out);
      else
        ln_fwd_k<T, NG><<<(unsigned)blocks, TPB, 0, s>>>(
            (const T*)x, (T*)y, w, b, mean, invstd, rows, d, eps);
    });
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
                 const void* dxadd, hipStream_t s) {
  int chunks = pdta_ln_chunks(rows);
  long rpc = (rows + chunks - 1) / chunks;
  int lds = 4 * 2 * d * sizeof(float);
  PDTA_DISPATCH_DTYPE16(dt, T, {
    PDTA_LN_DISPATCH_NG(d, NG, {
      if (dxadd)
        ln_bwd_k<T, NG, true><<<chunks, TPB, lds, s>>>(
            (const T*)x, (const T*)dy, (T*)dx, w, mean, invstd, part, rows,
            d, rpc, (const T*)dxadd);
      else
        ln_bwd_k<T, NG><<<chunks, TPB, lds, s>>>(
            (const T*)x, (const T*)dy, (T*)dx, w, mean, invstd, part, rows,
            d, rpc);
    });
  });
  // two-level finalize for long chunk lists (part carries 16 extra rows —
  // bind.cpp ln_bwd allocates chunks+16)
  constexpr int NSLAB = 16;
  if (chunks > 64) {
    float* p2 = part + (long)chunks * 2 * d;
    dim3 g1((d + 63) / 64, NSLAB);
    ln_bwd_psum_k<<<g1, 256, 0, s>>>(part, chunks, d, NSLAB, p2);
    ln_bwd_final_k<<<(d + 63) / 64, 256, 0, s>>>(p2, NSLAB, d, dw, db);
  } else {
    ln_bwd_final_k<<<(d + 63) / 64, 256, 0, s>>>(part, chunks, d, dw, db);
  }
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
