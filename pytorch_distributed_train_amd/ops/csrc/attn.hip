// Fused attention forward for the ViT path (SURVEY.md §2d softmax row;
// BASELINE config #5): one kernel computes P = softmax(scale · Q·K^T)
// (masked past nvalid — sequence padding to 256 stays exact, proven in
// tests/test_vit_cpu.py) and O = P·V for one (batch·head, 64-query) block.
//
// Replaces the scores-GEMM (fp32 [S,S] round trip through HBM), the
// standalone softmax, and the P·V GEMM of the decomposed path
// (bind.cpp attn_fwd): the scores never leave registers, P makes exactly
// one HBM write (kept for the decomposed backward), and K/V share one LDS
// slab (V is staged while softmax computes, hiding its latency).
//
// Zero-copy operands: q/k/v are element offsets into the interleaved
// [b, S, 3·h·hd] qkv projection (row stride 3·h·hd) — no permute/copy
// kernels anywhere, as in the decomposed path.
//
// Shape-specialized: head_dim == 64, sequence <= 256 (ViT-B/16); other
// shapes take the decomposed path. qkv/out/dout/dqkv may be UNPADDED
// (srows = actual rows per batch, e.g. 197): global row reads past
// nvalid-1 are clamped to the last valid row (the garbage they produce
// is annihilated — key cols >= nvalid are masked in the softmax, value
// rows >= nvalid carry zero probability, query rows >= nvalid are
// discarded at the store), row stores past nvalid write zeros, and the
// internal probs/ds buffers stay padded to SP so the dK/dV batched
// GEMMs can run K = SP against exactly-zero tail rows.
#include "common.h"
#include "mfma_tile.h"

namespace {

constexpr int SP = 256;  // padded sequence length
constexpr int HD = 64;   // head dim

template <typename T>
__global__ __launch_bounds__(256) void attn_fused_fwd_k(
    const T* __restrict__ qkv, T* __restrict__ probs, T* __restrict__ out,
    int h, int srows, int nvalid, float scale) {
  // grid: (SP/64 query blocks, b*h) — always the full padded range so
  // the probs tail rows get written (zeros). 4 waves; 16 query rows each.
  __shared__ __attribute__((aligned(16))) char qtile[64 * 128];
  __shared__ __attribute__((aligned(16))) char kv[SP * 128];  // K, then V
  __shared__ __attribute__((aligned(16))) char pimg[64 * 512];

  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int bh = blockIdx.y;
  int batch = bh / h, head = bh % h;
  long d = (long)h * HD, ld3 = 3 * d;
  const T* qbase = qkv + (long)batch * srows * ld3 + (long)head * HD;
  const T* kbase = qbase + d;
  const T* vbase = qbase + 2 * d;
  int q0 = blockIdx.x * 64;

  // ---- stage Q [64][128B] and K [256][128B] (glds: 1 KiB = 8 rows each);
  //      global rows >= nvalid clamp to the last valid row (see header)
  int row_in = lane >> 3, c16 = lane & 7;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int row = (wave * 2 + i) * 8 + row_in;
    int gr = q0 + row;
    if (gr >= nvalid) gr = nvalid - 1;
    const char* g = reinterpret_cast<const char*>(qbase + (long)gr * ld3) +
                    (c16 ^ (row & 7)) * 16;
    glds16(g, qtile + (wave * 2 + i) * 1024);
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int row = (wave * 8 + i) * 8 + row_in;
    int gr = row >= nvalid ? nvalid - 1 : row;
    const char* g =
        reinterpret_cast<const char*>(kbase + (long)gr * ld3) +
        (c16 ^ (row & 7)) * 16;
    glds16(g, kv + (wave * 8 + i) * 1024);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  // ---- S = Q·K^T for this wave's 16 query rows (all 256 key cols)
  f32x4 acc[16];
#pragma unroll
  for (int jn = 0; jn < 16; ++jn) acc[jn] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    s16x8 af =
        frag_read128(qtile, wave * 16 + (lane & 15), (lane >> 4) + kk * 4);
#pragma unroll
    for (int jn = 0; jn < 16; ++jn) {
      s16x8 bf =
          frag_read128(kv, jn * 16 + (lane & 15), (lane >> 4) + kk * 4);
      acc[jn] = MfmaOp<T>::run(af, bf, acc[jn]);
    }
  }
  __syncthreads();  // all waves done reading K: the slab becomes V

#pragma unroll
  for (int i = 0; i < 8; ++i) {  // stage V; its latency hides under softmax
    int row = (wave * 8 + i) * 8 + row_in;
    int gr = row >= nvalid ? nvalid - 1 : row;
    const char* g =
        reinterpret_cast<const char*>(vbase + (long)gr * ld3) +
        (c16 ^ (row & 7)) * 16;
    glds16(g, kv + (wave * 8 + i) * 1024);
  }

  // ---- masked softmax in registers (C layout: row (lane>>4)*4+r,
  //      col jn*16 + (lane&15); a row lives in the 16 lanes of one group)
  int colbase = lane & 15;
  float mx[4] = {-3.4e38f, -3.4e38f, -3.4e38f, -3.4e38f};
#pragma unroll
  for (int jn = 0; jn < 16; ++jn) {
    int col = jn * 16 + colbase;
    if (col < nvalid)
#pragma unroll
      for (int r = 0; r < 4; ++r) mx[r] = fmaxf(mx[r], acc[jn][r]);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
#pragma unroll
    for (int m = 1; m < 16; m <<= 1)
      mx[r] = fmaxf(mx[r], __shfl_xor(mx[r], m));
    mx[r] *= scale;
  }
  float sum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int jn = 0; jn < 16; ++jn) {
    int col = jn * 16 + colbase;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float e = (col < nvalid) ? __expf(acc[jn][r] * scale - mx[r]) : 0.f;
      acc[jn][r] = e;
      sum[r] += e;
    }
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
#pragma unroll
    for (int m = 1; m < 16; m <<= 1) sum[r] += __shfl_xor(sum[r], m);
    sum[r] = 1.f / sum[r];
  }

  // ---- write P: LDS A-image [64][512B] for the P·V pass + ONE global
  //      store (bf16/fp16) for the decomposed backward
  T* pg = probs + (long)bh * SP * SP + (long)q0 * SP;
#pragma unroll
  for (int jn = 0; jn < 16; ++jn) {
    int col = jn * 16 + colbase;
    int byte = col * 2;
    int chunk = byte >> 4, rem = byte & 15;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = wave * 16 + (lane >> 4) * 4 + r;
      // query rows past nvalid store ZERO probs — keeps the padded
      // probs/ds buffers' tail rows exactly zero so the K=SP dK/dV
      // GEMMs and the backward's P-tile reads annihilate clamp garbage
      T pv = q0 + row < nvalid ? from_f32<T>(acc[jn][r] * sum[r])
                               : from_f32<T>(0.f);
      *reinterpret_cast<T*>(pimg + row * 512 +
                            (((chunk ^ (row & 7)) << 4) | rem)) = pv;
      pg[(long)row * SP + col] = pv;
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // V staged
  __syncthreads();                                  // P image visible

  // ---- O = P·V (K-dim = 256 keys, 8 MFMA steps; V^T fragments via
  //      ds_read_b64_tr_b16 from the [k][m] V image)
  f32x4 acc2[4];
#pragma unroll
  for (int jn = 0; jn < 4; ++jn) acc2[jn] = {0.f, 0.f, 0.f, 0.f};
  unsigned vb = lds_addr_of(kv);
#pragma unroll
  for (int kt = 0; kt < 8; ++kt) {
    s16x8 af = frag_read_s<512>(pimg, wave * 16 + (lane & 15),
                                kt * 4 + (lane >> 4));
    unsigned long long lo[4], hi[4];
#pragma unroll
    for (int jn = 0; jn < 4; ++jn)
      frag_read_tr_issue(vb, kt * 32, jn * 16, lane, &lo[jn], &hi[jn]);
    tr_read_wait();
#pragma unroll
    for (int jn = 0; jn < 4; ++jn)
      acc2[jn] = MfmaOp<T>::run(af, tr_pair_to_frag(lo[jn], hi[jn]),
                                acc2[jn]);
  }

  T* obase = out + (long)batch * srows * d + (long)head * HD;
#pragma unroll
  for (int jn = 0; jn < 4; ++jn) {
    int col = jn * 16 + colbase;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = q0 + wave * 16 + (lane >> 4) * 4 + r;
      if (row < srows)
        obase[(long)row * d + col] =
            row < nvalid ? from_f32<T>(acc2[jn][r]) : from_f32<T>(0.f);
    }
  }
}

// Fused attention backward, stage 1: dP = dO·V^T,
// dS = scale · P ∘ (dP − rowsum(P ∘ dP)), and dQ = dS·K in one kernel —
// the [S,S] dP never touches HBM, dS goes straight from registers into
// the dQ MFMA pass (K reuses the V LDS slab), and dS is materialized
// once (bf16/fp16) for dK = dS^T·Q only.
template <typename T>
__global__ __launch_bounds__(256) void attn_fused_bwd_ds_k(
    const T* __restrict__ qkv, const T* __restrict__ p,
    const T* __restrict__ dout, T* __restrict__ ds, T* __restrict__ dqkv,
    int h, int srows, int nvalid, float scale) {
  __shared__ __attribute__((aligned(16))) char dotile[64 * 128];
  __shared__ __attribute__((aligned(16))) char vimg[SP * 128];
  __shared__ __attribute__((aligned(16))) char dsimg[64 * 512];

  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  int bh = blockIdx.y;
  int batch = bh / h, head = bh % h;
  long d = (long)h * HD, ld3 = 3 * d;
  const T* vbase =
      qkv + (long)batch * srows * ld3 + (long)head * HD + 2 * d;
  const T* kbase = vbase - d;  // k plane of the interleaved qkv
  const T* dobase = dout + (long)batch * srows * d + (long)head * HD;
  T* dqbase = dqkv + (long)batch * srows * ld3 + (long)head * HD;
  int q0 = blockIdx.x * 64;

  // row reads past nvalid-1 clamp to the last valid row; the garbage is
  // annihilated by the zero P rows/cols the forward stored (see header)
  int row_in = lane >> 3, c16 = lane & 7;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int row = (wave * 2 + i) * 8 + row_in;
    int gr = q0 + row;
    if (gr >= nvalid) gr = nvalid - 1;
    const char* g = reinterpret_cast<const char*>(dobase + (long)gr * d) +
                    (c16 ^ (row & 7)) * 16;
    glds16(g, dotile + (wave * 2 + i) * 1024);
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int row = (wave * 8 + i) * 8 + row_in;
    int gr = row >= nvalid ? nvalid - 1 : row;
    const char* g =
        reinterpret_cast<const char*>(vbase + (long)gr * ld3) +
        (c16 ^ (row & 7)) * 16;
    glds16(g, vimg + (wave * 8 + i) * 1024);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  // dP[m, n] = sum_f dO[m, f] · V[n, f]  (nt shape)
  f32x4 acc[16];
#pragma unroll
  for (int jn = 0; jn < 16; ++jn) acc[jn] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    s16x8 af =
        frag_read128(dotile, wave * 16 + (lane & 15), (lane >> 4) + kk * 4);
#pragma unroll
    for (int jn = 0; jn < 16; ++jn) {
      s16x8 bf =
          frag_read128(vimg, jn * 16 + (lane & 15), (lane >> 4) + kk * 4);
      acc[jn] = MfmaOp<T>::run(af, bf, acc[jn]);
    }
  }

  // stage this block's P tile [64][512B] into dsimg with glds16
  // (coalesced; the direct C-layout read was 64 scalar 2-B global loads
  // per lane), then read it back at the fragment layout from LDS. The
  // XOR swizzle matches the dS store below, so logical chunk c of row r
  // lives at LDS chunk c ^ (r & 7).
  const T* pg = p + (long)bh * SP * SP + (long)q0 * SP;
  T* dg = ds + (long)bh * SP * SP + (long)q0 * SP;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int pr = wave * 8 + i;               // row pair 0..31 (2 rows/call)
    int prow = pr * 2 + (lane >> 5);
    const char* g = reinterpret_cast<const char*>(pg + (long)prow * SP) +
                    (((lane & 31) ^ (prow & 7)) << 4);
    glds16(g, dsimg + pr * 1024);
  }
  // no barrier: each wave stages exactly the 16 rows it reads below
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  int colbase = lane & 15;
  float pv[16][4];
  float dot[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int jn = 0; jn < 16; ++jn) {
    int col = jn * 16 + colbase;
    int byte = col * 2;
    int chunk = byte >> 4, rem = byte & 15;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = wave * 16 + (lane >> 4) * 4 + r;
      pv[jn][r] = to_f32(*reinterpret_cast<const T*>(
          dsimg + row * 512 + (((chunk ^ (row & 7)) << 4) | rem)));
      dot[r] += pv[jn][r] * acc[jn][r];
    }
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
#pragma unroll
    for (int m = 1; m < 16; m <<= 1) dot[r] += __shfl_xor(dot[r], m);
  }
  __syncthreads();  // all waves done with V fragments: the slab becomes K
#pragma unroll
  for (int i = 0; i < 8; ++i) {  // stage K; hides under the dS writes
    int krow = (wave * 8 + i) * 8 + row_in;
    int gr = krow >= nvalid ? nvalid - 1 : krow;
    const char* g =
        reinterpret_cast<const char*>(kbase + (long)gr * ld3) +
        (c16 ^ (krow & 7)) * 16;
    glds16(g, vimg + (wave * 8 + i) * 1024);
  }
  // dS -> LDS A-image for dQ (global store is a vectorized flush below —
  // the per-element form was 64 scalar 2-B stores per lane)
#pragma unroll
  for (int jn = 0; jn < 16; ++jn) {
    int col = jn * 16 + colbase;
    int byte = col * 2;
    int chunk = byte >> 4, rem = byte & 15;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = wave * 16 + (lane >> 4) * 4 + r;
      T dsv = from_f32<T>(scale * pv[jn][r] * (acc[jn][r] - dot[r]));
      *reinterpret_cast<T*>(dsimg + row * 512 +
                            (((chunk ^ (row & 7)) << 4) | rem)) = dsv;
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // K staged
  __syncthreads();                                  // dS image visible
  // flush dS tile to global (consumed by dK = dS^T·Q): 16-B chunks,
  // un-swizzling the image rows
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int pr = wave * 8 + i;
    int prow = pr * 2 + (lane >> 5);
    int c = lane & 31;  // logical chunk
    uint4 v = *reinterpret_cast<const uint4*>(
        dsimg + prow * 512 + ((c ^ (prow & 7)) << 4));
    *reinterpret_cast<uint4*>(
        reinterpret_cast<char*>(dg + (long)prow * SP) + c * 16) = v;
  }

  // dQ = dS·K (K-dim = 256 keys; K^T fragments via ds_read_b64_tr_b16)
  f32x4 acc2[4];
#pragma unroll
  for (int jn = 0; jn < 4; ++jn) acc2[jn] = {0.f, 0.f, 0.f, 0.f};
  unsigned kb = lds_addr_of(vimg);
#pragma unroll
  for (int kt = 0; kt < 8; ++kt) {
    s16x8 af = frag_read_s<512>(dsimg, wave * 16 + (lane & 15),
                                kt * 4 + (lane >> 4));
    unsigned long long lo[4], hi[4];
#pragma unroll
    for (int jn = 0; jn < 4; ++jn)
      frag_read_tr_issue(kb, kt * 32, jn * 16, lane, &lo[jn], &hi[jn]);
    tr_read_wait();
#pragma unroll
    for (int jn = 0; jn < 4; ++jn)
      acc2[jn] = MfmaOp<T>::run(af, tr_pair_to_frag(lo[jn], hi[jn]),
                                acc2[jn]);
  }
#pragma unroll
  for (int jn = 0; jn < 4; ++jn) {
    int col = jn * 16 + colbase;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = q0 + wave * 16 + (lane >> 4) * 4 + r;
      if (row < srows)
        dqbase[(long)row * ld3 + col] =
            row < nvalid ? from_f32<T>(acc2[jn][r]) : from_f32<T>(0.f);
    }
  }
}

}  // namespace

extern "C" {

// 1 if the fused kernel handles this shape (bind.cpp attn_fwd dispatch).
// S is the ACTUAL per-batch row count (may be unpadded, e.g. 197); the
// kernel pads internally to SP.
int pdta_attn_fused_ok(int S, int hd) {
  return S <= SP && S >= 64 && hd == HD;
}

void pdta_attn_fused_fwd(const void* qkv, void* probs, void* out, int b,
                         int h, int srows, int nvalid, float scale, int dt,
                         hipStream_t s) {
  // always SP/64 blocks: the tail query blocks must WRITE the padded
  // probs rows (zeros) — the K=SP dK/dV GEMMs read all SP rows
  dim3 grid(SP / 64, b * h);
  PDTA_DISPATCH_DTYPE16(dt, T, {
    attn_fused_fwd_k<T><<<grid, 256, 0, s>>>(
        (const T*)qkv, (T*)probs, (T*)out, h, srows, nvalid, scale);
  });
}

void pdta_attn_fused_bwd_ds(const void* qkv, const void* p,
                            const void* dout, void* ds, void* dqkv, int b,
                            int h, int srows, int nvalid, float scale,
                            int dt, hipStream_t s) {
  dim3 grid(SP / 64, b * h);  // ds tail rows must be written (zeros) too
  PDTA_DISPATCH_DTYPE16(dt, T, {
    attn_fused_bwd_ds_k<T><<<grid, 256, 0, s>>>(
        (const T*)qkv, (const T*)p, (const T*)dout, (T*)ds, (T*)dqkv, h,
        srows, nvalid, scale);
  });
}

}  // extern "C"
