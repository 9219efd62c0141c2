// Fused SGD+momentum over the FLAT fp32 master/grad/momentum buffers
// (SURVEY.md §2d fused-SGD row; replaces torch _fused_sgd multi-tensor
// apply with one HBM-bound pass over one array), plus the AMP
// unscale+inf-check pass (grad_scaler.py:280 analog).
//
//   d = g*gs + wd*p;  m = mu*m + (1-damp)*d;  d' = nesterov ? d+mu*m : m;
//   p -= lr*d'
#include "common.h"

namespace {

constexpr int TPB = 256;

template <bool MOMENTUM, bool NESTEROV>
__global__ void fused_sgd_k(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, long n4, float lr,
                            float mu, float wd, float damp, float gs,
                            const float* __restrict__ lr_dev) {
  // lr from a device scalar when given: a hipGraph-captured launch then
  // honors later schedule updates (host lr would be baked into the graph)
  if (lr_dev) lr = *lr_dev;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) {
    float4 pv = reinterpret_cast<float4*>(p)[i];
    float4 gv = reinterpret_cast<const float4*>(g)[i];
    float4 mv;
    if (MOMENTUM) mv = reinterpret_cast<float4*>(m)[i];
    float* pp = &pv.x;
    const float* gp = &gv.x;
    float* mp = &mv.x;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float d = gp[j] * gs + wd * pp[j];
      if (MOMENTUM) {
        mp[j] = mu * mp[j] + (1.f - damp) * d;
        d = NESTEROV ? d + mu * mp[j] : mp[j];
      }
      pp[j] -= lr * d;
    }
    reinterpret_cast<float4*>(p)[i] = pv;
    if (MOMENTUM) reinterpret_cast<float4*>(m)[i] = mv;
  }
}

__global__ void unscale_check_k(float* __restrict__ g, long n4, float inv,
                                float* __restrict__ found) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  bool bad = false;
  for (; i < n4; i += stride) {
    float4 gv = reinterpret_cast<float4*>(g)[i];
    float* gp = &gv.x;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      gp[j] *= inv;
      bad |= !isfinite(gp[j]);
    }
    reinterpret_cast<float4*>(g)[i] = gv;
  }
  if (__any(bad)) {
    if ((threadIdx.x & 63) == 0) *found = 1.f;  // benign race: all write 1
  }
}

}  // namespace

extern "C" {

// n must be a multiple of 4 (the flat buffer is 64-element aligned).
void pdta_fused_sgd(float* p, const float* g, float* m, long n, float lr,
                    float mu, float wd, float damp, int nesterov, float gs,
                    const float* lr_dev, hipStream_t s) {
  long n4 = n / 4;
  int grid = ew_grid(n4, TPB, 1);
  if (mu != 0.f) {
    if (nesterov)
      fused_sgd_k<true, true><<<grid, TPB, 0, s>>>(p, g, m, n4, lr, mu, wd,
                                                   damp, gs, lr_dev);
    else
      fused_sgd_k<true, false><<<grid, TPB, 0, s>>>(p, g, m, n4, lr, mu, wd,
                                                    damp, gs, lr_dev);
  } else {
    fused_sgd_k<false, false><<<grid, TPB, 0, s>>>(p, g, nullptr, n4, lr, mu,
                                                   wd, damp, gs, lr_dev);
  }
}

void pdta_unscale_check(float* g, long n, float inv, float* found,
                        hipStream_t s) {
  long n4 = n / 4;
  unscale_check_k<<<ew_grid(n4, TPB, 1), TPB, 0, s>>>(g, n4, inv, found);
}

}  // extern "C"
