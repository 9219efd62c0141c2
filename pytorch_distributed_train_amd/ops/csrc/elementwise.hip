// Elementwise NHWC kernels: ReLU fwd, ReLU backward mask, residual
// add+ReLU (SURVEY.md §2d "ReLU / add-residual ... elementwise" row).
// Memory-bound: vectorized 8 elements/thread (16 B for bf16/fp16),
// grid-stride, fp32 math.
#include "common.h"

namespace {

template <typename T>
__global__ void relu_fwd_k(const T* __restrict__ x, T* __restrict__ y,
                           long n8) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n8; i += stride) {
    Vec8<T> v = load8(x + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      v.v[j] = from_f32<T>(fmaxf(to_f32(v.v[j]), 0.f));
    store8(y + i * 8, v);
  }
}

template <typename T>
__global__ void relu_bwd_mask_k(const T* __restrict__ dy,
                                const T* __restrict__ y, T* __restrict__ dx,
                                long n8) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n8; i += stride) {
    Vec8<T> g = load8(dy + i * 8);
    Vec8<T> v = load8(y + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      g.v[j] = to_f32(v.v[j]) > 0.f ? g.v[j] : from_f32<T>(0.f);
    store8(dx + i * 8, g);
  }
}

template <typename T>
__global__ void add_relu_k(const T* __restrict__ a, const T* __restrict__ b,
                           T* __restrict__ y, long n8) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n8; i += stride) {
    Vec8<T> va = load8(a + i * 8);
    Vec8<T> vb = load8(b + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      va.v[j] = from_f32<T>(fmaxf(to_f32(va.v[j]) + to_f32(vb.v[j]), 0.f));
    store8(y + i * 8, va);
  }
}

// scalar tail variants for n % 8 != 0 (never hit on NHWC activations where
// C % 64 == 0, but kept for generality)
template <typename T>
__global__ void relu_fwd_tail(const T* x, T* y, long start, long n) {
  long i = start + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) y[i] = from_f32<T>(fmaxf(to_f32(x[i]), 0.f));
}
template <typename T>
__global__ void relu_bwd_tail(const T* dy, const T* y, T* dx, long start,
                              long n) {
  long i = start + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) dx[i] = to_f32(y[i]) > 0.f ? dy[i] : from_f32<T>(0.f);
}
template <typename T>
__global__ void add_relu_tail(const T* a, const T* b, T* y, long start,
                              long n) {
  long i = start + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) y[i] = from_f32<T>(fmaxf(to_f32(a[i]) + to_f32(b[i]), 0.f));
}

constexpr int TPB = 256;

}  // namespace

extern "C" {

void pdta_relu_fwd(const void* x, void* y, long n, int dt, hipStream_t s) {
  long n8 = n / 8, tail = n % 8;
  PDTA_DISPATCH_DTYPE(dt, T, {
    if (n8)
      relu_fwd_k<T><<<ew_grid(n8, TPB, 1), TPB, 0, s>>>(
          (const T*)x, (T*)y, n8);
    if (tail)
      relu_fwd_tail<T><<<1, 64, 0, s>>>((const T*)x, (T*)y, n8 * 8, n);
  });
}

void pdta_relu_bwd_mask(const void* dy, const void* y, void* dx, long n,
                        int dt, hipStream_t s) {
  long n8 = n / 8, tail = n % 8;
  PDTA_DISPATCH_DTYPE(dt, T, {
    if (n8)
      relu_bwd_mask_k<T><<<ew_grid(n8, TPB, 1), TPB, 0, s>>>(
          (const T*)dy, (const T*)y, (T*)dx, n8);
    if (tail)
      relu_bwd_tail<T><<<1, 64, 0, s>>>((const T*)dy, (const T*)y, (T*)dx,
                                        n8 * 8, n);
  });
}

void pdta_add_relu(const void* a, const void* b, void* y, long n, int dt,
                   hipStream_t s) {
  long n8 = n / 8, tail = n % 8;
  PDTA_DISPATCH_DTYPE(dt, T, {
    if (n8)
      add_relu_k<T><<<ew_grid(n8, TPB, 1), TPB, 0, s>>>(
          (const T*)a, (const T*)b, (T*)y, n8);
    if (tail)
      add_relu_tail<T><<<1, 64, 0, s>>>((const T*)a, (const T*)b, (T*)y,
                                        n8 * 8, n);
  });
}

}  // extern "C"
