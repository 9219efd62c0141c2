// Fused softmax cross-entropy (SURVEY.md §2d softmax-CE row; replaces the
// fused log-softmax+NLL kernel behind nn.CrossEntropyLoss).
//
// fwd: one wave per row — lanes stride the class axis; wave-reduce max and
//      sum(exp); loss = lse - (1-eps)*l_y - eps*mean(l); saves lse fp32.
// bwd: dlogits = ((softmax - target) * scale * (*gout)) elementwise with
//      the saved lse; gout is a DEVICE scalar (no host sync in the loop).
#include "common.h"

namespace {

template <typename T>
__global__ void ce_fwd_k(const T* __restrict__ logits,
                         const long* __restrict__ labels,
                         float* __restrict__ loss, float* __restrict__ lse,
                         int B, int n, float eps) {
  int row = blockIdx.x * (blockDim.x / kWave) + (int)(threadIdx.x / kWave);
  if (row >= B) return;
  int lane = threadIdx.x & (kWave - 1);
  const T* lr = logits + (long)row * n;
  float mx = -3.4e38f;
  for (int c = lane; c < n; c += kWave) mx = fmaxf(mx, to_f32(lr[c]));
  mx = wave_reduce_max(mx);
  float sum = 0.f, lsum = 0.f;
  for (int c = lane; c < n; c += kWave) {
    float v = to_f32(lr[c]);
    sum += __expf(v - mx);
    lsum += v;
  }
  sum = wave_reduce_sum(sum);
  lsum = wave_reduce_sum(lsum);
  if (lane == 0) {
    float l = logf(sum) + mx;
    lse[row] = l;
    float ly = to_f32(lr[labels[row]]);
    float out = l - ly;
    if (eps > 0.f) out = (1.f - eps) * (l - ly) + eps * (l - lsum / n);
    loss[row] = out;
  }
}

template <typename T>
__global__ void ce_bwd_k(const T* __restrict__ logits,
                         const long* __restrict__ labels,
                         const float* __restrict__ lse,
                         const float* __restrict__ gout,
                         T* __restrict__ dlogits, int B, int n, float eps,
                         float scale) {
  float g = scale * *gout;
  long total = (long)B * n;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int row = (int)(i / n);
    int c = (int)(i % n);
    float p = __expf(to_f32(logits[i]) - lse[row]);
    float tgt = (labels[row] == c) ? (1.f - eps) : 0.f;
    if (eps > 0.f) tgt += eps / n;
    dlogits[i] = from_f32<T>((p - tgt) * g);
  }
}

}  // namespace

extern "C" {

void pdta_ce_fwd(const void* logits, const long* labels, float* loss,
                 float* lse, int B, int n, float eps, int dt,
                 hipStream_t s) {
  int waves_per_block = 4;  // 256 threads
  int grid = (B + waves_per_block - 1) / waves_per_block;
  PDTA_DISPATCH_DTYPE(dt, T, {
    ce_fwd_k<T><<<grid, waves_per_block * kWave, 0, s>>>(
        (const T*)logits, labels, loss, lse, B, n, eps);
  });
}

void pdta_ce_bwd(const void* logits, const long* labels, const float* lse,
                 const float* gout, void* dlogits, int B, int n, float eps,
                 float scale, int dt, hipStream_t s) {
  long total = (long)B * n;
  PDTA_DISPATCH_DTYPE(dt, T, {
    ce_bwd_k<T><<<ew_grid(total, 256, 1), 256, 0, s>>>(
        (const T*)logits, labels, lse, gout, (T*)dlogits, B, n, eps, scale);
  });
}

}  // extern "C"
