// Pooling kernels over [N,H,W,C] (SURVEY.md §2d maxpool/avgpool row).
//  - maxpool fwd: thread owns 8 channels at one output position, scans the
//    k*k window, records argmax as a u8 window offset per element;
//  - maxpool bwd: GATHER from the <=ceil(k/s)^2 covering outputs whose
//    argmax points back at this input element — no atomics, any dtype;
//  - global avgpool: fp32 row accumulation (H*W is small at this depth).
#include "common.h"

namespace {

constexpr int TPB = 256;

template <typename T>
__global__ void maxpool_fwd_k(const T* __restrict__ x, T* __restrict__ y,
                              uint8_t* __restrict__ idx, int N, int H, int W,
                              int C, int P, int Q, int K, int S, int pad) {
  // flat work item = ((n*P + p)*Q + q) * (C/8) + cg
  long total = (long)N * P * Q * (C / 8);
  long item = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; item < total; item += stride) {
    int cg = (int)(item % (C / 8));
    long t = item / (C / 8);
    int q = (int)(t % Q);
    t /= Q;
    int p = (int)(t % P);
    int n = (int)(t / P);
    int c0 = cg * 8;
    float best[8];
    uint8_t barg[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      best[j] = -3.4e38f;
      barg[j] = 0;
    }
    int h0 = p * S - pad, w0 = q * S - pad;
    for (int r = 0; r < K; ++r) {
      int h = h0 + r;
      if (h < 0 || h >= H) continue;
      for (int sCol = 0; sCol < K; ++sCol) {
        int w = w0 + sCol;
        if (w < 0 || w >= W) continue;
        Vec8<T> v = load8(x + (((long)n * H + h) * W + w) * C + c0);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = to_f32(v.v[j]);
          if (f > best[j]) {
            best[j] = f;
            barg[j] = (uint8_t)(r * K + sCol);
          }
        }
      }
    }
    Vec8<T> out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out.v[j] = from_f32<T>(best[j]);
    long o = (((long)n * P + p) * Q + q) * C + c0;
    store8(y + o, out);
    *reinterpret_cast<uint2*>(idx + o) = *reinterpret_cast<uint2*>(barg);
  }
}

template <typename T>
__global__ void maxpool_bwd_k(const T* __restrict__ dy,
                              const uint8_t* __restrict__ idx,
                              T* __restrict__ dx, int N, int H, int W, int C,
                              int P, int Q, int K, int S, int pad) {
  long total = (long)N * H * W * (C / 8);
  long item = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; item < total; item += stride) {
    int cg = (int)(item % (C / 8));
    long t = item / (C / 8);
    int w = (int)(t % W);
    t /= W;
    int h = (int)(t % H);
    int n = (int)(t / H);
    int c0 = cg * 8;
    float acc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] = 0.f;
    // outputs (p,q) whose window covers (h,w): p*S - pad <= h < p*S-pad+K
    int plo = (h + pad - K + S) / S;  // ceil((h+pad-K+1)/S) for h+pad-K+1>0
    if (plo < 0) plo = 0;
    int phi = (h + pad) / S;
    if (phi >= P) phi = P - 1;
    int qlo = (w + pad - K + S) / S;
    if (qlo < 0) qlo = 0;
    int qhi = (w + pad) / S;
    if (qhi >= Q) qhi = Q - 1;
    for (int p = plo; p <= phi; ++p) {
      int r = h - (p * S - pad);
      if (r < 0 || r >= K) continue;
      for (int q = qlo; q <= qhi; ++q) {
        int sCol = w - (q * S - pad);
        if (sCol < 0 || sCol >= K) continue;
        uint8_t want = (uint8_t)(r * K + sCol);
        long o = (((long)n * P + p) * Q + q) * C + c0;
        uint8_t args[8];
        *reinterpret_cast<uint2*>(args) =
            *reinterpret_cast<const uint2*>(idx + o);
        Vec8<T> g = load8(dy + o);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (args[j] == want) acc[j] += to_f32(g.v[j]);
      }
    }
    Vec8<T> out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out.v[j] = from_f32<T>(acc[j]);
    store8(dx + (((long)n * H + h) * W + w) * C + c0, out);
  }
}

template <typename T>
__global__ void gap_fwd_k(const T* __restrict__ x, T* __restrict__ y, int N,
                          long HW, int C) {
  // block.x covers channels (8 per thread), block.y = n
  int c8 = C / 8;
  int cpg = min(c8, TPB);
  int cg = blockIdx.x * cpg + (int)(threadIdx.x % cpg);
  if (cg >= c8 || threadIdx.x / cpg > 0) return;
  int n = blockIdx.y;
  int c0 = cg * 8;
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;
  const T* base = x + (long)n * HW * C + c0;
  for (long r = 0; r < HW; ++r) {
    Vec8<T> v = load8(base + r * C);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += to_f32(v.v[j]);
  }
  Vec8<T> out;
  float inv = 1.f / (float)HW;
#pragma unroll
  for (int j = 0; j < 8; ++j) out.v[j] = from_f32<T>(acc[j] * inv);
  store8(y + (long)n * C + c0, out);
}

template <typename T>
__global__ void gap_bwd_k(const T* __restrict__ dy, T* __restrict__ dx,
                          int N, long HW, int C) {
  long total = (long)N * HW * (C / 8);
  long item = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  float inv = 1.f / (float)HW;
  for (; item < total; item += stride) {
    int cg = (int)(item % (C / 8));
    long t = item / (C / 8);
    int n = (int)(t / HW);
    int c0 = cg * 8;
    Vec8<T> g = load8(dy + (long)n * C + c0);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      g.v[j] = from_f32<T>(to_f32(g.v[j]) * inv);
    store8(dx + item * 8, g);
  }
}

}  // namespace

extern "C" {

void pdta_maxpool_fwd(const void* x, void* y, uint8_t* idx, int N, int H,
                      int W, int C, int P, int Q, int K, int S, int pad,
                      int dt, hipStream_t s) {
  long total = (long)N * P * Q * (C / 8);
  PDTA_DISPATCH_DTYPE(dt, T, {
    maxpool_fwd_k<T><<<ew_grid(total, TPB, 1), TPB, 0, s>>>(
        (const T*)x, (T*)y, idx, N, H, W, C, P, Q, K, S, pad);
  });
}

void pdta_maxpool_bwd(const void* dy, const uint8_t* idx, void* dx, int N,
                      int H, int W, int C, int P, int Q, int K, int S,
                      int pad, int dt, hipStream_t s) {
  long total = (long)N * H * W * (C / 8);
  PDTA_DISPATCH_DTYPE(dt, T, {
    maxpool_bwd_k<T><<<ew_grid(total, TPB, 1), TPB, 0, s>>>(
        (const T*)dy, idx, (T*)dx, N, H, W, C, P, Q, K, S, pad);
  });
}

void pdta_gap_fwd(const void* x, void* y, int N, long HW, int C, int dt,
                  hipStream_t s) {
  int c8 = C / 8;
  int cpg = c8 < TPB ? c8 : TPB;
  dim3 g((c8 + cpg - 1) / cpg, N);
  PDTA_DISPATCH_DTYPE(dt, T, {
    gap_fwd_k<T><<<g, TPB, 0, s>>>((const T*)x, (T*)y, N, HW, C);
  });
}

void pdta_gap_bwd(const void* dy, void* dx, int N, long HW, int C, int dt,
                  hipStream_t s) {
  long total = (long)N * HW * (C / 8);
  PDTA_DISPATCH_DTYPE(dt, T, {
    gap_bwd_k<T><<<ew_grid(total, TPB, 1), TPB, 0, s>>>((const T*)dy, (T*)dx,
                                                        N, HW, C);
  });
}

}  // extern "C"
