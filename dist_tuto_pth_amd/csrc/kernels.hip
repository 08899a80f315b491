// kernels.hip — hand-written CDNA4 (gfx950/MI355X) kernels for every op
// on the reference training path plus the
// local-reduction primitives of the hand-rolled ring all-reduce (K14).
//
// The reference gets these ops implicitly from torch
// (Net.forward train_dist.py:64-71, SGD train_dist.py:110,124); here
// each is an explicit HIP kernel designed for CDNA4: 64-wide wavefronts,
// 256-thread blocks, vectorized float4 global access where layout
// permits, LDS staging for the conv/linear operand reuse, grid-stride
// loops capped so the 256-CU chip is filled without launch spam.
//
// fp32 throughout (the reference trains fp32); bf16 enters for the
// ring-allreduce reduction path (BASELINE config 5).
// Build: hipcc --offload-arch=gfx950 (build.py).  No CUDA paths.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_cooperative_groups.h>

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <cstdlib>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(cmd)                                                        \
  do {                                                                        \
    hipError_t e_ = (cmd);                                                    \
    if (e_ != hipSuccess)                                                     \
      throw std::runtime_error(std::string("HIP error: ") +                   \
                               hipGetErrorString(e_) + " @ " #cmd);           \
  } while (0)

static inline hipStream_t S(uintptr_t s) {
  return reinterpret_cast<hipStream_t>(s);
}

static inline int grid_for(int64_t work, int block, int per_thread = 1) {
  int64_t g = (work + (int64_t)block * per_thread - 1) /
              ((int64_t)block * per_thread);
  if (g > 4096) g = 4096;  // grid-stride beyond (G11: ~8 blocks/CU cap)
  if (g < 1) g = 1;
  return (int)g;
}

// ===========================================================================
// K1/K2 — direct convolution, stride 1, no padding (the only form Net
// uses: 5x5 kernels, C in {1,10}, K in {10,20}, H<=28).
// One block per batch element; the input plane (C*H*W <= 7.84 KB) and
// the full weight tensor (K*C*25 <= 19.5 KB) are staged in LDS, then
// the block's 256 threads sweep the K*OH*OW output points of that
// element.  Every input value is read from HBM exactly once per batch
// element regardless of K.
// ===========================================================================
__global__ void conv2d_fwd_kernel(const float* __restrict__ x,
                                  const float* __restrict__ w,
                                  const float* __restrict__ bias,
                                  float* __restrict__ out,
                                  int B, int C, int H, int W,
                                  int K, int R, int S_, int staged) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  const int OH = H - R + 1, OW = W - S_ + 1;
  // weights through LDS when they fit; global (L2-hot) otherwise
  const float* ws = w;
  if (staged) {
    const int wn = K * C * R * S_;
    for (int i = threadIdx.x; i < wn; i += blockDim.x) smem[i] = w[i];
    __syncthreads();
    ws = smem;
  }

  // one thread per output element across the whole tensor: fills the
  // 256-CU chip at any batch size (the per-batch-block form left half
  // the chip idle at B=128); x is tiny and L2-resident.
  const int64_t n_out = (int64_t)B * K * OH * OW;
  const int xn = C * H * W;
  const int on = K * OH * OW;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_out; i += (int64_t)gridDim.x * blockDim.x) {
    const int b = (int)(i / on);
    const int k = (int)((i / (OH * OW)) % K);
    const int oh = (int)((i / OW) % OH);
    const int ow = (int)(i % OW);
    float acc = bias ? bias[k] : 0.f;
    const float* xb = x + (int64_t)b * xn;
    const float* wk = ws + k * C * R * S_;
    for (int c = 0; c < C; ++c) {
      const float* xc = xb + c * H * W + oh * W + ow;
      const float* wc = wk + c * R * S_;
      #pragma unroll 5
      for (int r = 0; r < R; ++r) {
        const float* xrow = xc + r * W;
        const float* wrow = wc + r * S_;
        float a = 0.f;
        #pragma unroll 5
        for (int s = 0; s < S_; ++s) a += xrow[s] * wrow[s];
        acc += a;
      }
    }
    out[i] = acc;
  }
}

// backward dx: gx[b,c,h,w] = sum_k sum_{r,s} gy[b,k,h-r,w-s] * w[k,c,r,s]
// (valid range only).  gy plane (K*OH*OW <= 5.76 KB) + weights in LDS.
__global__ void conv2d_bwd_x_kernel(const float* __restrict__ gy,
                                    const float* __restrict__ w,
                                    float* __restrict__ gx,
                                    int B, int C, int H, int W,
                                    int K, int R, int S_, int staged) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  const int OH = H - R + 1, OW = W - S_ + 1;
  const float* ws = w;
  if (staged) {
    const int wn = K * C * R * S_;
    for (int i = threadIdx.x; i < wn; i += blockDim.x) smem[i] = w[i];
    __syncthreads();
    ws = smem;
  }

  const int64_t n_in = (int64_t)B * C * H * W;
  const int xn = C * H * W;
  const int gn = K * OH * OW;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_in; i += (int64_t)gridDim.x * blockDim.x) {
    const int b = (int)(i / xn);
    const int c = (int)((i / (H * W)) % C);
    const int h = (int)((i / W) % H);
    const int wcol = (int)(i % W);
    const float* gb = gy + (int64_t)b * gn;
    float acc = 0.f;
    const int r0 = max(0, h - OH + 1), r1 = min(R, h + 1);
    const int s0 = max(0, wcol - OW + 1), s1 = min(S_, wcol + 1);
    for (int k = 0; k < K; ++k) {
      const float* gk = gb + k * OH * OW;
      const float* wk = ws + (k * C + c) * R * S_;
      for (int r = r0; r < r1; ++r) {
        const int oh = h - r;
        const float* grow = gk + oh * OW + wcol;
        const float* wrow = wk + r * S_;
        float a = 0.f;
        for (int s = s0; s < s1; ++s) a += grow[-s] * wrow[s];
        acc += a;
      }
    }
    gx[i] = acc;
  }
}

// backward dw/db: per-block (per batch element) partials accumulated in
// LDS, then atomically added into gw/gb (gw zeroed by the caller).
__global__ void conv2d_bwd_w_kernel(const float* __restrict__ x,
                                    const float* __restrict__ gy,
                                    float* __restrict__ gw,
                                    float* __restrict__ gb,
                                    int B, int C, int H, int W,
                                    int K, int R, int S_, int bchunk) {
  const int OH = H - R + 1, OW = W - S_ + 1;
  const int xn = C * H * W;
  const int gn = K * OH * OW;
  const int wn = K * C * R * S_;
  const int nchunks = (B + bchunk - 1) / bchunk;
  const int chunk = blockIdx.y;
  const int b0 = chunk * bchunk;
  const int b1 = min(B, b0 + bchunk);

  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < wn;
       i += gridDim.x * blockDim.x) {
    const int k = i / (C * R * S_);
    const int c = (i / (R * S_)) % C;
    const int r = (i / S_) % R;
    const int sx = i % S_;
    float acc = 0.f;
    for (int b = b0; b < b1; ++b) {
      const float* gk = gy + (int64_t)b * gn + k * OH * OW;
      const float* xc = x + (int64_t)b * xn + c * H * W + r * W + sx;
      for (int oh = 0; oh < OH; ++oh) {
        const float* grow = gk + oh * OW;
        const float* xrow = xc + oh * W;
        float a = 0.f;
        for (int ow = 0; ow < OW; ++ow) a += grow[ow] * xrow[ow];
        acc += a;
      }
    }
    if (nchunks == 1) gw[i] = acc;
    else atomicAdd(&gw[i], acc);
  }
  if (gb) {
    for (int k = blockIdx.x * blockDim.x + threadIdx.x; k < K;
         k += gridDim.x * blockDim.x) {
      float acc = 0.f;
      for (int b = b0; b < b1; ++b) {
        const float* gk = gy + (int64_t)b * gn + k * OH * OW;
        for (int i2 = 0; i2 < OH * OW; ++i2) acc += gk[i2];
      }
      if (nchunks == 1) gb[k] = acc;
      else atomicAdd(&gb[k], acc);
    }
  }
}

// ===========================================================================
// K3+K4 — fused 2x2/stride-2 maxpool + ReLU (train_dist.py:65-66).
// Forward stores the winning index (0..3) packed with the sign; the
// backward writes all four window slots (pool windows are disjoint), so
// gx needs no pre-zeroing pass.
// ===========================================================================
__global__ void maxpool2d_relu_fwd_kernel(const float* __restrict__ x,
                                          float* __restrict__ out,
                                          int* __restrict__ idx,
                                          int64_t planes, int H, int W) {
  const int OH = H / 2, OW = W / 2;
  const int64_t n = planes * OH * OW;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t p = i / (OH * OW);
    const int oh = (int)((i / OW) % OH);
    const int ow = (int)(i % OW);
    const float* xp = x + p * H * W + (oh * 2) * W + ow * 2;
    float v0 = xp[0], v1 = xp[1], v2 = xp[W], v3 = xp[W + 1];
    int am = 0;
    float m = v0;
    if (v1 > m) { m = v1; am = 1; }
    if (v2 > m) { m = v2; am = 2; }
    if (v3 > m) { m = v3; am = 3; }
    out[i] = m > 0.f ? m : 0.f;
    idx[i] = m > 0.f ? am : (am | 4);  // bit2: ReLU clipped
  }
}

__global__ void maxpool2d_relu_bwd_kernel(const float* __restrict__ gy,
                                          const int* __restrict__ idx,
                                          float* __restrict__ gx,
                                          int64_t planes, int H, int W) {
  const int OH = H / 2, OW = W / 2;
  const int64_t n = planes * OH * OW;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t p = i / (OH * OW);
    const int oh = (int)((i / OW) % OH);
    const int ow = (int)(i % OW);
    float* gp = gx + p * H * W + (oh * 2) * W + ow * 2;
    const int v = idx[i];
    const int am = v & 3;
    const float g = (v & 4) ? 0.f : gy[i];
    gp[0] = am == 0 ? g : 0.f;
    gp[1] = am == 1 ? g : 0.f;
    gp[W] = am == 2 ? g : 0.f;
    gp[W + 1] = am == 3 ? g : 0.f;
  }
}

// ===========================================================================
// K4 — standalone ReLU (elementwise, float4-vectorized)
// ===========================================================================
__global__ void relu_fwd_kernel(const float* __restrict__ x,
                                float* __restrict__ out, int64_t n) {
  const int64_t n4 = n / 4;
  const float4* x4 = reinterpret_cast<const float4*>(x);
  float4* o4 = reinterpret_cast<float4*>(out);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (int64_t)gridDim.x * blockDim.x) {
    float4 v = x4[i];
    v.x = v.x > 0.f ? v.x : 0.f;
    v.y = v.y > 0.f ? v.y : 0.f;
    v.z = v.z > 0.f ? v.z : 0.f;
    v.w = v.w > 0.f ? v.w : 0.f;
    o4[i] = v;
  }
  for (int64_t i = n4 * 4 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = x[i] > 0.f ? x[i] : 0.f;
}

__global__ void relu_bwd_kernel(const float* __restrict__ gy,
                                const float* __restrict__ out,
                                float* __restrict__ gx, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    gx[i] = out[i] > 0.f ? gy[i] : 0.f;
}

// ===========================================================================
// K5/K6 — dropout.  Counter-based hash RNG (wang/xxhash-style mix of
// (seed, index)): stateless, reproducible from the host-passed seed,
// no RNG-state tensor.  Channelwise variant draws one number per
// (batch, channel) plane (Dropout2d, train_dist.py:60).
// ===========================================================================
// device-side seed bump: lets dropout RNG advance across hipGraph
// replays (the seed lives in device memory; one tiny kernel increments
// it before each dropout draw, stream-ordered).
__global__ void bump_seed_kernel(unsigned long long* s) {
  if (threadIdx.x == 0 && blockIdx.x == 0)
    *s += 0x9E3779B97F4A7C15ull;
}

// fused step prologue: advance the dropout seed (when training) and
// zero the loss accumulator in ONE tiny launch instead of a bump
// kernel + a hipMemsetAsync fill kernel (each ~4.5 us of pure launch
// cost at the reference's batch sizes).
__global__ void step_prologue_kernel(unsigned long long* s, float* loss) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    if (s) *s += 0x9E3779B97F4A7C15ull;
    *loss = 0.f;
  }
}

__device__ inline uint32_t mix32(uint64_t seed, uint64_t idx) {
  uint64_t z = seed + 0x9E3779B97F4A7C15ull * (idx + 1);
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return (uint32_t)((z ^ (z >> 31)) >> 16);
}

__global__ void dropout_fwd_kernel(const float* __restrict__ x,
                                   float* __restrict__ out,
                                   uint8_t* __restrict__ mask, int64_t n,
                                   float p, float scale,
                                   const unsigned long long* __restrict__ sp) {
  const uint32_t thresh = (uint32_t)(p * 4294967296.0);
  const uint64_t seed = sp[0];
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t r = mix32(seed, (uint64_t)i) << 16 | (mix32(seed ^ 0xabcd, i) & 0xffff);
    uint8_t keep = r >= thresh;
    mask[i] = keep;
    out[i] = keep ? x[i] * scale : 0.f;
  }
}

__global__ void dropout_bwd_kernel(const float* __restrict__ gy,
                                   const uint8_t* __restrict__ mask,
                                   float* __restrict__ gx, int64_t n,
                                   float scale) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    gx[i] = mask[i] ? gy[i] * scale : 0.f;
}

__global__ void dropout2d_fwd_kernel(const float* __restrict__ x,
                                     float* __restrict__ out,
                                     uint8_t* __restrict__ mask,
                                     int64_t planes, int64_t hw, float p,
                                     float scale,
                                     const unsigned long long* __restrict__ sp) {
  const uint32_t thresh = (uint32_t)(p * 4294967296.0);
  const uint64_t seed = sp[0];
  const int64_t n = planes * hw;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t pl = i / hw;
    uint32_t r = mix32(seed, (uint64_t)pl) << 16 |
                 (mix32(seed ^ 0xabcd, pl) & 0xffff);
    uint8_t keep = r >= thresh;
    if (i % hw == 0) mask[pl] = keep;
    out[i] = keep ? x[i] * scale : 0.f;
  }
}

__global__ void dropout2d_bwd_kernel(const float* __restrict__ gy,
                                     const uint8_t* __restrict__ mask,
                                     float* __restrict__ gx,
                                     int64_t planes, int64_t hw,
                                     float scale) {
  const int64_t n = planes * hw;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    gx[i] = mask[i / hw] ? gy[i] * scale : 0.f;
}

// ===========================================================================
// K7/K8 — linear: out[B,N] = x[B,K] @ w[N,K]^T + bias, optional fused
// ReLU epilogue (fc1, train_dist.py:68).  Net shapes (K=320,N=50 and
// K=50,N=10) are tiny: the whole weight panel fits LDS, each block
// stages w once and sweeps a stripe of batch rows; x rows are read
// once, each thread owning one (b,n) output.
// ===========================================================================
__global__ void linear_fwd_kernel(const float* __restrict__ x,
                                  const float* __restrict__ w,
                                  const float* __restrict__ bias,
                                  float* __restrict__ out,
                                  int B, int K, int N, int fuse_relu,
                                  int staged) {
  // weights staged through LDS when they fit (Net's fc shapes);
  // read from global (L2-resident, shared by every block) otherwise
  extern __shared__ __attribute__((aligned(16))) float smem[];
  const float* ws = w;
  if (staged) {
    const int wn = N * K;
    for (int i = threadIdx.x; i < wn; i += blockDim.x) smem[i] = w[i];
    __syncthreads();
    ws = smem;
  }

  const int64_t n_out = (int64_t)B * N;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_out; i += (int64_t)gridDim.x * blockDim.x) {
    const int b = (int)(i / N);
    const int n = (int)(i % N);
    const float* xr = x + (int64_t)b * K;
    const float* wr = ws + n * K;
    float acc = bias ? bias[n] : 0.f;
    int k = 0;
    for (; k + 4 <= K; k += 4)
      acc += xr[k] * wr[k] + xr[k + 1] * wr[k + 1] +
             xr[k + 2] * wr[k + 2] + xr[k + 3] * wr[k + 3];
    for (; k < K; ++k) acc += xr[k] * wr[k];
    if (fuse_relu && acc < 0.f) acc = 0.f;
    out[i] = acc;
  }
}

// gx[B,K] = gy'[B,N] @ w[N,K]  (gy' = gy masked by out>0 when fused)
__global__ void linear_bwd_x_kernel(const float* __restrict__ gy,
                                    const float* __restrict__ out,
                                    const float* __restrict__ w,
                                    float* __restrict__ gx,
                                    int B, int K, int N, int staged) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  const float* ws = w;
  if (staged) {
    const int wn = N * K;
    for (int i = threadIdx.x; i < wn; i += blockDim.x) smem[i] = w[i];
    __syncthreads();
    ws = smem;
  }

  const int64_t n_out = (int64_t)B * K;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_out; i += (int64_t)gridDim.x * blockDim.x) {
    const int b = (int)(i / K);
    const int k = (int)(i % K);
    const float* gr = gy + (int64_t)b * N;
    const float* orow = out ? out + (int64_t)b * N : nullptr;
    float acc = 0.f;
    for (int n = 0; n < N; ++n) {
      float g = gr[n];
      if (orow && orow[n] <= 0.f) g = 0.f;
      acc += g * ws[n * K + k];
    }
    gx[i] = acc;
  }
}

// gw[N,K] += sum_b gy'[b,n] * x[b,k] ; gb[N] += sum_b gy'[b,n]
// Each block owns a batch stripe, accumulates partials, atomically adds.
__global__ void linear_bwd_w_kernel(const float* __restrict__ x,
                                    const float* __restrict__ gy,
                                    const float* __restrict__ out,
                                    float* __restrict__ gw,
                                    float* __restrict__ gb,
                                    int B, int K, int N, int bchunk) {
  const int nchunks = (B + bchunk - 1) / bchunk;
  const int b0 = blockIdx.y * bchunk;
  const int b1 = min(B, b0 + bchunk);
  const int wn = N * K;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < wn;
       i += gridDim.x * blockDim.x) {
    const int n = i / K;
    const int k = i % K;
    float acc = 0.f;
    for (int b = b0; b < b1; ++b) {
      float g = gy[(int64_t)b * N + n];
      if (out && out[(int64_t)b * N + n] <= 0.f) g = 0.f;
      acc += g * x[(int64_t)b * K + k];
    }
    if (nchunks == 1) gw[i] = acc;
    else atomicAdd(&gw[i], acc);
  }
  if (gb) {
    for (int n = blockIdx.x * blockDim.x + threadIdx.x; n < N;
         n += gridDim.x * blockDim.x) {
      float acc = 0.f;
      for (int b = b0; b < b1; ++b) {
        float g = gy[(int64_t)b * N + n];
        if (out && out[(int64_t)b * N + n] <= 0.f) g = 0.f;
        acc += g;
      }
      if (nchunks == 1) gb[n] = acc;
      else atomicAdd(&gb[n], acc);
    }
  }
}

// ===========================================================================
// K9/K10 — log_softmax over dim 1 (train_dist.py:71) and NLL loss
// (train_dist.py:120), plus the fused single-pass form.
// N is tiny (10): one thread per row, serial max/sum over N — the
// tensor is L2-resident at these sizes and the op is launch-bound.
// ===========================================================================
__global__ void log_softmax_fwd_kernel(const float* __restrict__ x,
                                       float* __restrict__ out,
                                       int B, int N) {
  for (int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += (int64_t)gridDim.x * blockDim.x) {
    const float* xr = x + b * N;
    float* orow = out + b * N;
    float m = xr[0];
    for (int i = 1; i < N; ++i) m = fmaxf(m, xr[i]);
    float s = 0.f;
    for (int i = 0; i < N; ++i) s += __expf(xr[i] - m);
    const float lse = m + __logf(s);
    for (int i = 0; i < N; ++i) orow[i] = xr[i] - lse;
  }
}

// gx = gy - exp(out) * sum(gy)
__global__ void log_softmax_bwd_kernel(const float* __restrict__ gy,
                                       const float* __restrict__ out,
                                       float* __restrict__ gx,
                                       int B, int N) {
  for (int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += (int64_t)gridDim.x * blockDim.x) {
    const float* gr = gy + b * N;
    const float* orow = out + b * N;
    float s = 0.f;
    for (int i = 0; i < N; ++i) s += gr[i];
    for (int i = 0; i < N; ++i)
      gx[b * N + i] = gr[i] - __expf(orow[i]) * s;
  }
}

__global__ void nll_loss_fwd_kernel(const float* __restrict__ logp,
                                    const int64_t* __restrict__ target,
                                    float* __restrict__ loss, int B, int N) {
  __shared__ float part[256];
  float acc = 0.f;
  for (int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += (int64_t)gridDim.x * blockDim.x)
    acc -= logp[b * N + target[b]];
  part[threadIdx.x] = acc;
  __syncthreads();
  for (int s = blockDim.x / 2; s > 0; s >>= 1) {
    if (threadIdx.x < s) part[threadIdx.x] += part[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(loss, part[0] / B);
}

// gloss arrives as a DEVICE scalar pointer so the whole backward is
// hipGraph-capturable (no host read of the grad seed).
__global__ void nll_loss_bwd_kernel(const int64_t* __restrict__ target,
                                    float* __restrict__ gx,
                                    const float* __restrict__ gloss,
                                    int B, int N) {
  const float sc = -gloss[0] / B;
  for (int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += (int64_t)gridDim.x * blockDim.x)
    gx[b * N + target[b]] = sc;
}

// fused: logp + mean NLL in one pass (K9+K10)
__global__ void log_softmax_nll_fwd_kernel(const float* __restrict__ x,
                                           const int64_t* __restrict__ tgt,
                                           float* __restrict__ logp,
                                           float* __restrict__ loss,
                                           int B, int N) {
  __shared__ float part[256];
  float acc = 0.f;
  for (int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += (int64_t)gridDim.x * blockDim.x) {
    const float* xr = x + b * N;
    float* lr = logp + b * N;
    float m = xr[0];
    for (int i = 1; i < N; ++i) m = fmaxf(m, xr[i]);
    float s = 0.f;
    for (int i = 0; i < N; ++i) s += __expf(xr[i] - m);
    const float lse = m + __logf(s);
    for (int i = 0; i < N; ++i) lr[i] = xr[i] - lse;
    acc -= lr[tgt[b]];
  }
  part[threadIdx.x] = acc;
  __syncthreads();
  for (int s2 = blockDim.x / 2; s2 > 0; s2 >>= 1) {
    if (threadIdx.x < s2) part[threadIdx.x] += part[threadIdx.x + s2];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(loss, part[0] / B);
}

// gx = (exp(logp) - onehot) * gloss / B
__global__ void log_softmax_nll_bwd_kernel(const float* __restrict__ logp,
                                           const int64_t* __restrict__ tgt,
                                           float* __restrict__ gx,
                                           const float* __restrict__ gloss,
                                           int B, int N) {
  const float sc = gloss[0] / B;
  for (int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += (int64_t)gridDim.x * blockDim.x) {
    const float* lr = logp + b * N;
    float* gr = gx + b * N;
    const int64_t t = tgt[b];
    for (int i = 0; i < N; ++i)
      gr[i] = (__expf(lr[i]) - (i == t ? 1.f : 0.f)) * sc;
  }
}

// ===========================================================================
// K11-K13 — fused multi-tensor SGD+momentum: buf = mu*buf + g;
// p -= lr*buf; optionally g = 0 (K12 folded in).  All of Net's 8
// tensors in ONE launch: pointer table passed by value.
// ===========================================================================
#define MT_MAX 32
struct MtArgs {
  float* p[MT_MAX];
  float* g[MT_MAX];
  float* buf[MT_MAX];
  int64_t numel[MT_MAX];
  int64_t offset[MT_MAX];  // prefix sum for flat indexing
  int count;
  int64_t total;
};

__global__ void sgd_step_kernel(MtArgs a, float lr, float mu,
                                int zero_grad) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < a.total; i += (int64_t)gridDim.x * blockDim.x) {
    // binary search the tensor containing flat index i
    int lo = 0, hi = a.count - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (i >= a.offset[mid]) lo = mid; else hi = mid - 1;
    }
    const int64_t j = i - a.offset[lo];
    float g = a.g[lo][j];
    if (a.buf[lo]) {
      g = mu * a.buf[lo][j] + g;
      a.buf[lo][j] = g;
    }
    a.p[lo][j] -= lr * g;
    if (zero_grad) a.g[lo][j] = 0.f;
  }
}

// ===========================================================================
// K14 helpers — local reductions for the hand-rolled / hand-tuned ring
// all-reduce (allreduce.py:26,31 corrected): dst += src, vectorized.
// fp32 and bf16 (BASELINE config 5).
//
// MFMA considered and rejected for this op: a column reduction is
// out[j] = sum_p src[p][j], i.e. ones[1xP] x src[PxN].  On a 16x16x4
// f32 MFMA every C row would hold the same 4-row partial sum, so one
// 4-cycle instruction reduces 4x16 = 64 elements = 16 elem/cycle/wave,
// while plain VALU v_add_f32 reduces 64 elem/cycle/wave — the matrix
// pipe wastes a 16x output replication on a rank-1 operand.  The op is
// HBM-bandwidth-bound anyway (reads P*N floats once); the float4
// grid-stride form below saturates that.  MFMA belongs to GEMM-shaped
// work, which this framework's models reach through the register-
// blocked conv kernels above (shapes are 5x5/latency-bound, below the
// MFMA payoff threshold) and hipBLASLt/MIOpen for ResNet-50.
// ===========================================================================
__global__ void add_inplace_f32_kernel(float* __restrict__ dst,
                                       const float* __restrict__ src,
                                       int64_t n) {
  const int64_t n4 = n / 4;
  float4* d4 = reinterpret_cast<float4*>(dst);
  const float4* s4 = reinterpret_cast<const float4*>(src);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (int64_t)gridDim.x * blockDim.x) {
    float4 a = d4[i], b = s4[i];
    a.x += b.x; a.y += b.y; a.z += b.z; a.w += b.w;
    d4[i] = a;
  }
  for (int64_t i = n4 * 4 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] += src[i];
}

__global__ void add_inplace_bf16_kernel(__hip_bfloat16* __restrict__ dst,
                                        const __hip_bfloat16* __restrict__ src,
                                        int64_t n) {
  // 8 bf16 (16 B) per lane (G13): short4-shaped access via uint4
  const int64_t n8 = n / 8;
  uint4* d8 = reinterpret_cast<uint4*>(dst);
  const uint4* s8 = reinterpret_cast<const uint4*>(src);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint4 a = d8[i], b = s8[i];
    __hip_bfloat162* ah = reinterpret_cast<__hip_bfloat162*>(&a);
    const __hip_bfloat162* bh = reinterpret_cast<const __hip_bfloat162*>(&b);
    #pragma unroll
    for (int k = 0; k < 4; ++k) ah[k] = __hadd2(ah[k], bh[k]);
    d8[i] = a;
  }
  for (int64_t i = n8 * 8 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] = __hadd(dst[i], src[i]);
}

// column reduction for the full-mesh reduce-scatter (K14): after the
// grouped p2p exchange, rank i holds P-1 peer chunks contiguously in
// scratch; one kernel folds them all into the owned chunk.
// dst[j] (+)= sum_p src[p*stride + j].  fp32 accumulates fp32; bf16
// accumulates in fp32 and rounds ONCE at the end (better than serial
// bf16 adds — BASELINE config 5 accuracy note, SURVEY.md §7 hard part d).
__global__ void reduce_columns_f32_kernel(float* __restrict__ dst,
                                          const float* __restrict__ src,
                                          int P, int64_t stride,
                                          int64_t n, float scale) {
  const int64_t n4 = n / 4;
  float4* d4 = reinterpret_cast<float4*>(dst);
  const float4* s4 = reinterpret_cast<const float4*>(src);
  const int64_t stride4 = stride / 4;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (int64_t)gridDim.x * blockDim.x) {
    float4 a = d4[i];
    for (int p = 0; p < P; ++p) {
      float4 b = s4[p * stride4 + i];
      a.x += b.x; a.y += b.y; a.z += b.z; a.w += b.w;
    }
    a.x *= scale; a.y *= scale; a.z *= scale; a.w *= scale;
    d4[i] = a;
  }
  for (int64_t i = n4 * 4 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float a = dst[i];
    for (int p = 0; p < P; ++p) a += src[p * stride + i];
    dst[i] = a * scale;
  }
}

__global__ void reduce_columns_bf16_kernel(
    __hip_bfloat16* __restrict__ dst,
    const __hip_bfloat16* __restrict__ src, int P, int64_t stride,
    int64_t n, float scale) {
  const int64_t n8 = n / 8;
  uint4* d8 = reinterpret_cast<uint4*>(dst);
  const uint4* s8 = reinterpret_cast<const uint4*>(src);
  const int64_t stride8 = stride / 8;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint4 av = d8[i];
    const __hip_bfloat162* ah = reinterpret_cast<__hip_bfloat162*>(&av);
    float2 acc[4];
    #pragma unroll
    for (int k = 0; k < 4; ++k) acc[k] = __bfloat1622float2(ah[k]);
    for (int p = 0; p < P; ++p) {
      uint4 bv = s8[p * stride8 + i];
      const __hip_bfloat162* bh =
          reinterpret_cast<const __hip_bfloat162*>(&bv);
      #pragma unroll
      for (int k = 0; k < 4; ++k) {
        float2 b = __bfloat1622float2(bh[k]);
        acc[k].x += b.x;
        acc[k].y += b.y;
      }
    }
    uint4 ov;
    __hip_bfloat162* oh = reinterpret_cast<__hip_bfloat162*>(&ov);
    #pragma unroll
    for (int k = 0; k < 4; ++k)
      oh[k] = __float22bfloat162_rn(
          make_float2(acc[k].x * scale, acc[k].y * scale));
    d8[i] = ov;
  }
  for (int64_t i = n8 * 8 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float a = __bfloat162float(dst[i]);
    for (int p = 0; p < P; ++p)
      a += __bfloat162float(src[p * stride + i]);
    dst[i] = __float2bfloat16(a * scale);
  }
}

__global__ void scale_f32_kernel(float* __restrict__ dst, float s,
                                 int64_t n) {
  const int64_t n4 = n / 4;
  float4* d4 = reinterpret_cast<float4*>(dst);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (int64_t)gridDim.x * blockDim.x) {
    float4 a = d4[i];
    a.x *= s; a.y *= s; a.z *= s; a.w *= s;
    d4[i] = a;
  }
  for (int64_t i = n4 * 4 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] *= s;
}

// grid-throttled float4 copy: a stand-in for a link-bound transfer in
// single-GPU overlap probes.  RCCL moves an xGMI peer copy with a few
// workgroups at ~153 GB/s per link — a full-rate DtoD memcpy (5+ TB/s,
// all of HBM) is the WRONG model for it.  Limiting the grid caps the
// copy's HBM draw so the concurrent reduce kernel has headroom, which
// is exactly the real pipeline's situation.
__global__ void copy_throttled_f32_kernel(float* __restrict__ dst,
                                          const float* __restrict__ src,
                                          int64_t n) {
  const int64_t n4 = n / 4;
  float4* d4 = reinterpret_cast<float4*>(dst);
  const float4* s4 = reinterpret_cast<const float4*>(src);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (int64_t)gridDim.x * blockDim.x)
    d4[i] = s4[i];
  for (int64_t i = n4 * 4 + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] = src[i];
}

// ===========================================================================
// Fused whole-Net forward / backward (the launch-bound-regime lever).
//
// At the reference's batch sizes the per-op pipeline is bound by ~30
// kernel launches of ~5 us each, not by arithmetic (rocprof profile,
// profiles/).  These two kernels run the ENTIRE Net forward
// (train_dist.py:64-71: conv1 -> pool+relu -> conv2 -> dropout2d ->
// pool+relu -> fc1+relu -> dropout -> fc2 -> log_softmax -> NLL) and
// the entire data backward in ONE kernel each, one workgroup per batch
// element, activations staged through LDS; only the small per-batch
// weight-gradient reductions stay as separate (chunked) kernels.
// Architecture constants are Net's (SURVEY.md §2.1 'Net'): they are
// hard-coded, which is the point — this is the flagship model's fast
// path; the modular kernels above remain the general path.
// ===========================================================================
#define N_C1K 10        // conv1 out channels
#define N_C2K 20        // conv2 out channels
#define N_A1 (10*24*24) // conv1 out 5760
#define N_P1 (10*12*12) // pool1 out 1440
#define N_A2 (20*8*8)   // conv2 out 1280
#define N_P2 320        // pool2 out (= fc1 in)
#define N_H1 50         // fc1 out
#define N_CLS 10        // classes

// partial-row access, parameterized on coherence: the in-launch fold
// variant stores partials sc1 (device-coherent, straight to the
// coherence point) and reads them back sc1 — ZERO fences, so the
// concurrently-running tile blocks keep their L1/L2 activation
// locality.  The first fold attempt used the plain-store + agent
// release/acquire recipe and measured 23-28% SLOWER end to end: one
// `buffer_wbl2` L2 writeback per block x 1744 blocks evicted the very
// activations the partial blocks re-read (ledger).  Partial traffic is
// ~1.8 MB/step, so the slower sc1 store path is immaterial.
template <bool SC1>
__device__ __forceinline__ void gw_st(float* p, float v) {
  if (SC1)
    __hip_atomic_store(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  else
    *p = v;
}
template <bool SC1>
__device__ __forceinline__ float gw_ld(const float* p) {
  if (SC1)
    return __hip_atomic_load(p, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
  return *p;
}

// One sample's full forward (conv1..log_softmax+NLL) by one 256-thread
// workgroup, activations staged through caller-provided LDS buffers.
// Returns -— on tid 0 only — the sample's log-prob at the target
// (callers accumulate the NLL loss); other lanes return 0.
// Shared by net_fused_fwd_kernel and the single-launch net_step_kernel.
// SPLIT mode (fused fwd at small B): TWO sibling workgroups per sample
// fill the otherwise half-idle chip (B=128 launches 128 blocks on 256
// CUs).  Both stage x/w1 and compute conv1+pool1 fully (duplicating
// the cheap stage costs less than a second exchange), each computes
// HALF of conv2's 20 output channels (and stages only its half of w2),
// sibling 1 publishes its p2 half with sc1 stores + a per-sample flag
// and exits; sibling 0 consumes it and runs fc1/fc2/softmax alone.
// If the sibling is not yet resident (co-residency is not contractual)
// sibling 0 falls back to computing the other half itself from its own
// LDS p1 — identical values, so the duplicate global stores are benign
// and the wait can never deadlock.
template <bool SPLIT = false>
__device__ __forceinline__ float net_fwd_sample(
    int b, int tid, int B, int training, uint64_t seed,
    int half, unsigned int* __restrict__ flag, unsigned int token,
    const float* __restrict__ x,
    const float* __restrict__ w1, const float* __restrict__ b1,
    const float* __restrict__ w2, const float* __restrict__ b2,
    const float* __restrict__ wf1, const float* __restrict__ bf1,
    const float* __restrict__ wf2, const float* __restrict__ bf2,
    const int64_t* __restrict__ tgt,
    float* __restrict__ p1_ws, uint8_t* __restrict__ idx1_ws,
    uint8_t* __restrict__ m2_ws, float* __restrict__ p2_ws,
    uint8_t* __restrict__ idx2_ws, float* __restrict__ h1_ws,
    uint8_t* __restrict__ m3_ws, float* __restrict__ d3_ws,
    float* __restrict__ logp_ws,
    // LDS carve (sizes: 784, 260, 1440, 5020, 320, 50, 10)
    float* xs, float* w1s, float* p1, float* w2s,
    float* p2, float* d3, float* logits) {
  float ret = 0.f;
  {
    // stage input + conv weights
    for (int i = tid; i < 784; i += 256) xs[i] = x[(int64_t)b * 784 + i];
    for (int i = tid; i < N_C1K * 25; i += 256) w1s[i] = w1[i];
    if (tid < N_C1K) w1s[N_C1K * 25 + tid] = b1[tid];
    {
      const int wlo = SPLIT ? half * (N_C2K / 2) * 250 : 0;
      const int whi = SPLIT ? wlo + (N_C2K / 2) * 250 : N_C2K * 250;
      for (int i = wlo + tid; i < whi; i += 256) w2s[i] = w2[i];
      const int klo = SPLIT ? half * (N_C2K / 2) : 0;
      const int khi = SPLIT ? klo + N_C2K / 2 : N_C2K;
      if (klo + tid < khi) w2s[N_C2K * 250 + klo + tid] = b2[klo + tid];
    }
    __syncthreads();

    // conv1 + pool1 + relu in ONE register-blocked stage: each thread
    // owns a 2x2 pooling window, computes its four conv outputs in
    // four independent accumulator chains (the kernel is LDS-latency
    // bound at ~1 wave/SIMD — profiles/), then pools in registers.
    // The 5760-float conv1 activation never touches LDS.
    for (int i = tid; i < N_P1; i += 256) {
      const int c = i / 144, ph = (i / 12) % 12, pw = i % 12;
      const float* wk = w1s + c * 25;
      const float* xp = xs + ph * 2 * 28 + pw * 2;
      const float bias = w1s[N_C1K * 25 + c];
      float q00 = bias, q01 = bias, q10 = bias, q11 = bias;
      #pragma unroll
      for (int r = 0; r < 5; ++r) {
        #pragma unroll
        for (int s = 0; s < 5; ++s) {
          const float w = wk[r * 5 + s];
          const float* xr = xp + r * 28 + s;
          q00 += xr[0] * w;
          q01 += xr[1] * w;
          q10 += xr[28] * w;
          q11 += xr[29] * w;
        }
      }
      int am = 0; float m = q00;
      if (q01 > m) { m = q01; am = 1; }
      if (q10 > m) { m = q10; am = 2; }
      if (q11 > m) { m = q11; am = 3; }
      const float o = m > 0.f ? m : 0.f;
      p1[i] = o;
      if (!SPLIT || half == 0) {
        p1_ws[(int64_t)b * N_P1 + i] = o;
        idx1_ws[(int64_t)b * N_P1 + i] =
            (uint8_t)(m > 0.f ? am : (am | 4));
      }
    }
    __syncthreads();

    // conv2 + dropout2d + pool2 + relu, same register-blocked shape:
    // one thread per pooled cell, four conv outputs in four chains,
    // channel dropout applied in registers before the max.
    const int c2lo = SPLIT ? half * (N_P2 / 2) : 0;
    const int c2hi = SPLIT ? c2lo + N_P2 / 2 : N_P2;
    for (int i = c2lo + tid; i < c2hi; i += 256) {
      const int k = i / 16, ph = (i / 4) % 4, pw = i % 4;
      const float bias = w2s[N_C2K * 250 + k];
      float q00 = bias, q01 = bias, q10 = bias, q11 = bias;
      #pragma unroll
      for (int c = 0; c < 10; ++c) {
        const float* pp = p1 + c * 144 + ph * 2 * 12 + pw * 2;
        const float* wc = w2s + (k * 10 + c) * 25;
        #pragma unroll
        for (int r = 0; r < 5; ++r) {
          #pragma unroll
          for (int s = 0; s < 5; ++s) {
            const float w = wc[r * 5 + s];
            const float* pr = pp + r * 12 + s;
            q00 += pr[0] * w;
            q01 += pr[1] * w;
            q10 += pr[12] * w;
            q11 += pr[13] * w;
          }
        }
      }
      if (training) {
        const uint32_t rr = mix32(seed, (uint64_t)b * N_C2K + k);
        const float dsc = (rr >= 0x80000000u) ? 2.f : 0.f;
        q00 *= dsc; q01 *= dsc; q10 *= dsc; q11 *= dsc;
      }
      int am = 0; float m = q00;
      if (q01 > m) { m = q01; am = 1; }
      if (q10 > m) { m = q10; am = 2; }
      if (q11 > m) { m = q11; am = 3; }
      const float o = m > 0.f ? m : 0.f;
      p2[i] = o;
      gw_st<SPLIT>(p2_ws + (int64_t)b * N_P2 + i, o);
      idx2_ws[(int64_t)b * N_P2 + i] = (uint8_t)(m > 0.f ? am : (am | 4));
    }
    if ((!SPLIT || half == 0) && training && tid < N_C2K) {
      const uint32_t rr = mix32(seed, (uint64_t)b * N_C2K + tid);
      m2_ws[(int64_t)b * N_C2K + tid] = rr >= 0x80000000u;
    }
    __syncthreads();
    if (SPLIT && half == 1) {
      // publish: p2 half is sc1-stored (at the coherence point once
      // vmcnt retires) — drain, then write this STEP's token.  A
      // token (not a 0/1 flag) makes stale publishes harmless: if a
      // fallback fired last step, this late write still never matches
      // a later step's expected token, so no reset is ever needed.
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
      if (tid == 0)
        __hip_atomic_store(flag, token, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
      return 0.f;  // fc/softmax belong to sibling 0
    }
    if (SPLIT && half == 0) {
      // consume the sibling's half: bounded poll for THIS step's
      // token, then sc1 loads into our LDS p2.  On timeout compute
      // the half ourselves (fallback — co-residency is the norm but
      // not contractual).
      __shared__ unsigned s_got;
      if (tid == 0) {
        unsigned got = 0;
        for (int it = 0; it < 60000; ++it) {
          if (__hip_atomic_load(flag, __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_AGENT) == token) {
            got = 1u;
            break;
          }
          __builtin_amdgcn_s_sleep(4);
        }
        if (got)  // consume: under hipGraph replay the token repeats,
                  // so the next replay must wait for a fresh publish
          __hip_atomic_store(flag, 0u, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
        s_got = got;
      }
      __syncthreads();
      if (s_got) {
        for (int i = N_P2 / 2 + tid; i < N_P2; i += 256)
          p2[i] = gw_ld<true>(p2_ws + (int64_t)b * N_P2 + i);
      } else {  // sibling never ran: do its channels from our own p1
        for (int i = N_P2 / 2 + tid; i < N_P2; i += 256) {
          const int k = i / 16, ph = (i / 4) % 4, pw = i % 4;
          float q00 = b2[k], q01 = q00, q10 = q00, q11 = q00;
          #pragma unroll
          for (int c = 0; c < 10; ++c) {
            const float* pp = p1 + c * 144 + ph * 2 * 12 + pw * 2;
            const float* wc = w2 + (k * 10 + c) * 25;  // global: our
            #pragma unroll                             // w2s half only
            for (int r = 0; r < 5; ++r) {
              #pragma unroll
              for (int s = 0; s < 5; ++s) {
                const float w = wc[r * 5 + s];
                const float* pr = pp + r * 12 + s;
                q00 += pr[0] * w;
                q01 += pr[1] * w;
                q10 += pr[12] * w;
                q11 += pr[13] * w;
              }
            }
          }
          if (training) {
            const uint32_t rr = mix32(seed, (uint64_t)b * N_C2K + k);
            const float dsc = (rr >= 0x80000000u) ? 2.f : 0.f;
            q00 *= dsc; q01 *= dsc; q10 *= dsc; q11 *= dsc;
          }
          float m = fmaxf(fmaxf(q00, q01), fmaxf(q10, q11));
          p2[i] = m > 0.f ? m : 0.f;
        }
      }
      __syncthreads();
    }

    // fc1 (320->50) + relu + dropout: 4 threads per output, shfl reduce
    if (tid < N_H1 * 4) {
      const int n = tid >> 2, q = tid & 3;
      const float* wr = wf1 + n * N_P2 + q * 80;
      const float* pr = p2 + q * 80;
      float e0 = 0.f, e1 = 0.f;
      #pragma unroll 4
      for (int k = 0; k < 80; k += 2) {
        e0 += pr[k] * wr[k];
        e1 += pr[k + 1] * wr[k + 1];
      }
      float acc = e0 + e1;
      acc += __shfl_down(acc, 1, 4);
      acc += __shfl_down(acc, 2, 4);
      if (q == 0) {
        float h = acc + bf1[n];
        h = h > 0.f ? h : 0.f;
        h1_ws[(int64_t)b * N_H1 + n] = h;
        float dd = h;
        uint8_t keep = 1;
        if (training) {
          const uint32_t rr = mix32(seed ^ 0x5bd1e995u,
                                    (uint64_t)b * N_H1 + n);
          keep = rr >= 0x80000000u;
          dd = keep ? h * 2.f : 0.f;
        }
        m3_ws[(int64_t)b * N_H1 + n] = keep;
        d3_ws[(int64_t)b * N_H1 + n] = dd;
        d3[n] = dd;
      }
    }
    __syncthreads();

    // fc2 (50->10)
    if (tid < N_CLS) {
      const float* wr = wf2 + tid * N_H1;
      float acc = bf2[tid];
      #pragma unroll 10
      for (int k = 0; k < N_H1; ++k) acc += d3[k] * wr[k];
      logits[tid] = acc;
    }
    __syncthreads();

    // log_softmax + NLL (one lane)
    if (tid == 0) {
      float m = logits[0];
      #pragma unroll
      for (int i = 1; i < N_CLS; ++i) m = fmaxf(m, logits[i]);
      float ssum = 0.f;
      #pragma unroll
      for (int i = 0; i < N_CLS; ++i) ssum += __expf(logits[i] - m);
      const float lse = m + __logf(ssum);
      float lp_t = 0.f;
      const int64_t t = tgt[b];
      #pragma unroll
      for (int i = 0; i < N_CLS; ++i) {
        const float lp = logits[i] - lse;
        logp_ws[(int64_t)b * N_CLS + i] = lp;
        if (i == (int)t) lp_t = lp;
      }
      ret = lp_t;
    }
    __syncthreads();
  }
  return ret;
}

__global__ void
__launch_bounds__(256)
net_fused_fwd_kernel(
    const float* __restrict__ x,      // [B,1,28,28]
    const float* __restrict__ w1, const float* __restrict__ b1,
    const float* __restrict__ w2, const float* __restrict__ b2,
    const float* __restrict__ wf1, const float* __restrict__ bf1,
    const float* __restrict__ wf2, const float* __restrict__ bf2,
    const int64_t* __restrict__ tgt,
    float* __restrict__ p1_ws,        // [B,1440]
    uint8_t* __restrict__ idx1_ws,    // [B,1440]
    uint8_t* __restrict__ m2_ws,      // [B,20]
    float* __restrict__ p2_ws,        // [B,320]
    uint8_t* __restrict__ idx2_ws,    // [B,320]
    float* __restrict__ h1_ws,        // [B,50]
    uint8_t* __restrict__ m3_ws,      // [B,50]
    float* __restrict__ d3_ws,        // [B,50]
    float* __restrict__ logp_ws,      // [B,10]
    float* __restrict__ loss,         // scalar (pre-zeroed) — legacy mode
    float* __restrict__ loss_part,    // [gridDim.x] or null: when set,
                                      // per-block partials (no atomics,
                                      // no pre-zeroing; the combine
                                      // kernel finalizes the loss)
    const unsigned long long* __restrict__ seed_p,
    int B, int training) {
  __shared__ __attribute__((aligned(16))) float xs[784];
  __shared__ float w1s[N_C1K * 25 + N_C1K];
  __shared__ float p1[N_P1];
  __shared__ float w2s[N_C2K * 10 * 25 + N_C2K];
  __shared__ float p2[N_P2];
  __shared__ float d3[N_H1];
  __shared__ float logits[N_CLS];
  const int tid = threadIdx.x;
  const uint64_t seed = seed_p[0];

  float lsum = 0.f;
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const float lp_t = net_fwd_sample(
        b, tid, B, training, seed, 0, nullptr, 0u, x, w1, b1, w2, b2,
        wf1,
        bf1, wf2, bf2, tgt, p1_ws, idx1_ws, m2_ws, p2_ws, idx2_ws,
        h1_ws, m3_ws, d3_ws, logp_ws, xs, w1s, p1, w2s, p2, d3, logits);
    if (tid == 0) lsum += -lp_t / B;
  }
  if (tid == 0) {
    if (loss_part) loss_part[blockIdx.x] = lsum;
    else atomicAdd(loss, lsum);
  }
}

// Two sibling workgroups per sample (small-B fused fwd: B=128 alone
// leaves half the 256 CUs idle).  blockIdx.x = 2*b + half; sibling 1
// computes conv2 channels 10..19, publishes its p2 half (sc1 + flag)
// and exits; sibling 0 runs the rest.  Grid is exactly 2*B.
__global__ void
__launch_bounds__(256)
net_fused_fwd_split_kernel(
    const float* __restrict__ x,
    const float* __restrict__ w1, const float* __restrict__ b1,
    const float* __restrict__ w2, const float* __restrict__ b2,
    const float* __restrict__ wf1, const float* __restrict__ bf1,
    const float* __restrict__ wf2, const float* __restrict__ bf2,
    const int64_t* __restrict__ tgt,
    float* __restrict__ p1_ws, uint8_t* __restrict__ idx1_ws,
    uint8_t* __restrict__ m2_ws, float* __restrict__ p2_ws,
    uint8_t* __restrict__ idx2_ws, float* __restrict__ h1_ws,
    uint8_t* __restrict__ m3_ws, float* __restrict__ d3_ws,
    float* __restrict__ logp_ws, float* __restrict__ loss,
    float* __restrict__ loss_part,
    const unsigned long long* __restrict__ seed_p,
    int B, int training, unsigned int* __restrict__ flags,
    unsigned int token) {
  __shared__ __attribute__((aligned(16))) float xs[784];
  __shared__ float w1s[N_C1K * 25 + N_C1K];
  __shared__ float p1[N_P1];
  __shared__ float w2s[N_C2K * 10 * 25 + N_C2K];
  __shared__ float p2[N_P2];
  __shared__ float d3[N_H1];
  __shared__ float logits[N_CLS];
  const int tid = threadIdx.x;
  const uint64_t seed = seed_p[0];
  const int b = blockIdx.x >> 1, half = blockIdx.x & 1;
  const float lp_t = net_fwd_sample<true>(
      b, tid, B, training, seed, half, flags + b, token, x, w1, b1, w2,
      b2, wf1, bf1, wf2, bf2, tgt, p1_ws, idx1_ws, m2_ws, p2_ws,
      idx2_ws, h1_ws, m3_ws, d3_ws, logp_ws, xs, w1s, p1, w2s, p2, d3,
      logits);
  if (tid == 0) {
    const float lsum = half == 0 ? -lp_t / B : 0.f;
    if (loss_part) loss_part[blockIdx.x] = lsum;
    else if (half == 0) atomicAdd(loss, lsum);
  }
}

// One (sample, sibling-half)'s data backward: loss grad -> g_a1 (conv1
// output grad).  `split` sibling workgroups cooperate on each sample:
// the cheap early stages (g_logits .. gd2p staging, ~35k FMA) are
// recomputed by every sibling, the hot conv2-bwd_x stage (720k FMA) is
// partitioned by pooled position.  w2s must be pre-staged with conv2's
// weights (5000 floats) by the caller.  Shared by net_fused_bwd_kernel
// and the single-launch net_step_kernel.
__device__ __forceinline__ void net_bwd_sample(
    int b, int half, int split, int tid, int B, int training, float sc,
    const float* __restrict__ wf1, const float* __restrict__ wf2,
    const int64_t* __restrict__ tgt,
    const uint8_t* __restrict__ idx1_ws,
    const uint8_t* __restrict__ m2_ws,
    const uint8_t* __restrict__ idx2_ws,
    const float* __restrict__ h1_ws,
    const uint8_t* __restrict__ m3_ws,
    const float* __restrict__ logp_ws,
    float* __restrict__ glog_ws, float* __restrict__ gh1_ws,
    float* __restrict__ ga2_ws, float* __restrict__ ga1_ws,
    // LDS carve (sizes: 5000, 10, 50, 50, 320, 6080)
    const float* w2s, float* glg, float* gd3, float* gh1, float* gp2,
    float* gd2p) {
  {
    // g_logits = (exp(logp) - onehot) * gl / B
    __syncthreads();
    if (tid < N_CLS) {
      const float lp = logp_ws[(int64_t)b * N_CLS + tid];
      const float g = (__expf(lp) -
                       (tid == (int)tgt[b] ? 1.f : 0.f)) * sc;
      glg[tid] = g;
      if (half == 0)
        glog_ws[(int64_t)b * N_CLS + tid] = g;
    }
    __syncthreads();

    // g_d3 = wf2^T g_logits ; through dropout + relu -> g_h1pre
    if (tid < N_H1) {
      float acc = 0.f;
      #pragma unroll
      for (int n = 0; n < N_CLS; ++n)
        acc += wf2[n * N_H1 + tid] * glg[n];
      gd3[tid] = acc;
      float g = acc;
      if (training) {
        g = m3_ws[(int64_t)b * N_H1 + tid] ? g * 2.f : 0.f;
      }
      if (h1_ws[(int64_t)b * N_H1 + tid] <= 0.f) g = 0.f;
      gh1[tid] = g;
      if (half == 0)
        gh1_ws[(int64_t)b * N_H1 + tid] = g;
    }
    __syncthreads();

    // g_p2 = wf1^T g_h1pre  (320 outputs x 50), four accumulator
    // chains over the reduction dim (wf1 rows are coalesced global
    // reads; the single 50-deep load+FMA chain was latency-bound)
    for (int i = tid; i < N_P2; i += 256) {
      float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
      #pragma unroll
      for (int n = 0; n < 48; n += 4) {
        a0 += wf1[n * N_P2 + i] * gh1[n];
        a1 += wf1[(n + 1) * N_P2 + i] * gh1[n + 1];
        a2 += wf1[(n + 2) * N_P2 + i] * gh1[n + 2];
        a3 += wf1[(n + 3) * N_P2 + i] * gh1[n + 3];
      }
      a0 += wf1[48 * N_P2 + i] * gh1[48];
      a1 += wf1[49 * N_P2 + i] * gh1[49];
      gp2[i] = (a0 + a1) + (a2 + a3);
    }
    __syncthreads();

    // pool2 bwd + dropout2d bwd -> g_a2, scattered STRAIGHT into the
    // padded gd2p tile with dropout applied (r2: the old path staged
    // through a dense gd2 buffer — zero + scatter + full re-read, an
    // extra 2.5k LDS ops per sample for nothing; gd2 is gone)
    for (int i = tid; i < N_C2K * 304; i += 256) gd2p[i] = 0.f;
    __syncthreads();
    for (int i = tid; i < N_P2; i += 256) {
      const int c = i / 16, oh = (i / 4) % 4, ow = i % 4;
      const uint8_t v = idx2_ws[(int64_t)b * N_P2 + i];
      if (!(v & 4)) {
        const int am = v & 3;
        const int fr = oh * 2 + (am >> 1), fc = ow * 2 + (am & 1);
        float g = gp2[i];
        if (training)
          g = m2_ws[(int64_t)b * N_C2K + c] ? g * 2.f : 0.f;
        gd2p[c * 304 + (fr + 4) * 19 + (fc + 4)] = g;
      }
    }
    __syncthreads();
    if (half == 0) {  // stash the dense conv2-out grad for the gw pass
      for (int i = tid; i < N_A2; i += 256) {
        const int k = i / 64, oh = (i / 8) % 8, ow = i % 8;
        ga2_ws[(int64_t)b * N_A2 + i] =
            gd2p[k * 304 + (oh + 4) * 19 + (ow + 4)];
      }
    }
    __syncthreads();

    // conv2 bwd_x through the padded tile + pool1 bwd, register-blocked
    // 4-wide: each thread owns FOUR horizontally-adjacent pooled
    // positions (one accumulator chain each); a (k, r) row contributes
    // 8 consecutive gd2p values shared by all four outputs, cutting
    // LDS reads ~2.5x vs one-output-per-thread (the kernel is
    // LDS-latency-bound at this occupancy — profiles/).  Each pooled
    // position owns its 2x2 ga1 window exclusively, so all four slots
    // are written with no zeroing pass.  Partitioned across the
    // `split` sibling workgroups (360 strips divisible by 1/2/4/8).
    const int seg = (N_P1 / 4) / split;
    const int t_end = (half + 1) * seg;
    for (int t = half * seg + tid; t < t_end; t += 256) {
      const int c = t / 36, rem = t % 36;
      const int h = rem / 3, wc0 = (rem % 3) * 4;
      float q0 = 0.f, q1 = 0.f, q2 = 0.f, q3 = 0.f;
      for (int k = 0; k < N_C2K; ++k) {
        const float* gk = gd2p + k * 304;
        const float* wk = w2s + (k * 10 + c) * 25;
        #pragma unroll
        for (int r = 0; r < 5; ++r) {
          const float* row = gk + (h - r + 4) * 19 + wc0;
          const float w0 = wk[r * 5 + 0], w1 = wk[r * 5 + 1];
          const float w2v = wk[r * 5 + 2], w3 = wk[r * 5 + 3];
          const float w4 = wk[r * 5 + 4];
          const float e0 = row[0], e1 = row[1], e2 = row[2];
          const float e3 = row[3], e4 = row[4], e5 = row[5];
          const float e6 = row[6], e7 = row[7];
          q0 += e4 * w0 + e3 * w1 + e2 * w2v + e1 * w3 + e0 * w4;
          q1 += e5 * w0 + e4 * w1 + e3 * w2v + e2 * w3 + e1 * w4;
          q2 += e6 * w0 + e5 * w1 + e4 * w2v + e3 * w3 + e2 * w4;
          q3 += e7 * w0 + e6 * w1 + e5 * w2v + e4 * w3 + e3 * w4;
        }
      }
      const float qq[4] = {q0, q1, q2, q3};
      const int64_t ib = (int64_t)b * N_P1 + c * 144 + h * 12 + wc0;
      float* gp_base = ga1_ws + (int64_t)b * N_A1 + c * 576 +
                       h * 2 * 24 + wc0 * 2;
      #pragma unroll
      for (int j = 0; j < 4; ++j) {
        const uint8_t v = idx1_ws[ib + j];
        const int am = v & 3;
        const float g = (v & 4) ? 0.f : qq[j];
        float* gp = gp_base + j * 2;
        gp[0] = am == 0 ? g : 0.f;
        gp[1] = am == 1 ? g : 0.f;
        gp[24] = am == 2 ? g : 0.f;
        gp[25] = am == 3 ? g : 0.f;
      }
    }
    __syncthreads();
  }
}

// Fused data-gradient backward: from loss grad to g_a1 (conv1 output
// grad) in one kernel; weight gradients are reduced afterwards by the
// chunked partial/combine kernels over the stashes.
__global__ void
__launch_bounds__(256)
net_fused_bwd_kernel(
    const float* __restrict__ w2, const float* __restrict__ wf1,
    const float* __restrict__ wf2,
    const int64_t* __restrict__ tgt,
    const float* __restrict__ gl,      // dLoss (device scalar)
    const uint8_t* __restrict__ idx1_ws,
    const uint8_t* __restrict__ m2_ws,
    const uint8_t* __restrict__ idx2_ws,
    const float* __restrict__ h1_ws,
    const uint8_t* __restrict__ m3_ws,
    const float* __restrict__ logp_ws,
    float* __restrict__ glog_ws,       // [B,10]  (fc2 out grad)
    float* __restrict__ gh1_ws,        // [B,50]  (fc1 pre-relu grad)
    float* __restrict__ ga2_ws,        // [B,1280] (conv2 out grad)
    float* __restrict__ ga1_ws,        // [B,5760] (conv1 out grad)
    int B, int training, int split) {
  __shared__ __attribute__((aligned(16))) float w2s[N_C2K * 250];
  __shared__ float glg[N_CLS];
  __shared__ float gd3[N_H1];
  __shared__ float gh1[N_H1];
  __shared__ float gp2[N_P2];
  // zero-padded conv2-out grad [k][16][19]: entry (k,oh+4,ow+4) holds
  // the conv2-out grad at [k][oh][ow]; the pad makes the transposed-
  // conv window reads branch-free so the 500-FMA loop unrolls with
  // ILP.  Row stride 19 (not 16): PMC measured 59.8% LDSBankConflict
  // at B=4096 from the power-of-2 stride (profiles/pmc_r2.md); over
  // the loop's (h, wc0) lane pattern a brute-force scan shows stride
  // 19 is conflict-FREE on 64 banks (17 still left 3-way clusters).
  __shared__ float gd2p[N_C2K * 304];
  const int tid = threadIdx.x;

  for (int i = tid; i < N_C2K * 250; i += 256) w2s[i] = w2[i];
  const float sc = gl[0] / B;

  for (int bb = blockIdx.x; bb < B * split; bb += gridDim.x) {
    net_bwd_sample(bb / split, bb % split, split, tid, B, training, sc,
                   wf1, wf2, tgt, idx1_ws, m2_ws, idx2_ws, h1_ws, m3_ws,
                   logp_ws, glog_ws, gh1_ws, ga2_ws, ga1_ws,
                   w2s, glg, gd3, gh1, gp2, gd2p);
  }
}

// Combined forward + data-backward in ONE kernel (round-2 experiment,
// VERDICT r1 #9).  In the training loop the loss gradient is the
// constant 1, so sample b's data backward depends ONLY on sample b's
// forward — both run in the same workgroup with a __syncthreads where
// the inter-kernel dependency used to be.  Saves one ~4.5 us dispatch
// and the w2 re-staging (5000 floats/WG) the separate bwd kernel pays.
// Cost: the backward loses its `split` sibling-workgroup parallelism
// (grid = B, not B*split), so at small B the chip is underfilled.
// The backward LDS carve aliases the forward's dead buffers: total
// 5020 (w2s, shared by both phases — same [k*10+c]*25 weight layout)
// + 7168 floats = 47.7 KB/WG -> 3 WGs/CU.
__global__ void
__launch_bounds__(256)
net_fused_fwdbwd_kernel(
    const float* __restrict__ x,
    const float* __restrict__ w1, const float* __restrict__ b1,
    const float* __restrict__ w2, const float* __restrict__ b2,
    const float* __restrict__ wf1, const float* __restrict__ bf1,
    const float* __restrict__ wf2, const float* __restrict__ bf2,
    const int64_t* __restrict__ tgt,
    float* __restrict__ p1_ws, uint8_t* __restrict__ idx1_ws,
    uint8_t* __restrict__ m2_ws, float* __restrict__ p2_ws,
    uint8_t* __restrict__ idx2_ws, float* __restrict__ h1_ws,
    uint8_t* __restrict__ m3_ws, float* __restrict__ d3_ws,
    float* __restrict__ logp_ws,
    float* __restrict__ glog_ws, float* __restrict__ gh1_ws,
    float* __restrict__ ga2_ws, float* __restrict__ ga1_ws,
    float* __restrict__ loss_part,
    const unsigned long long* __restrict__ seed_p, int B, int training) {
  __shared__ __attribute__((aligned(16))) float w2s[N_C2K * 250 + N_C2K];
  __shared__ __attribute__((aligned(16))) float pool[7168];
  const int tid = threadIdx.x;
  const uint64_t seed = seed_p[0];
  // forward carve
  float* xs = pool;              // 784
  float* w1s = pool + 784;       // 260
  float* p1 = pool + 1048;       // 1440  (784+260=1044 -> pad to 1048)
  float* p2 = pool + 2488;       // 320
  float* d3 = pool + 2808;       // 50
  float* logits = pool + 2860;   // 10
  // backward carve (aliased onto the forward's dead buffers)
  float* gd2p = pool;            // 6080 (= 20 k-planes x 16 x 19)
  float* gp2 = pool + 6080;      // 320
  float* glg = pool + 6400;      // 10
  float* gd3 = pool + 6412;      // 50
  float* gh1 = pool + 6464;      // 50 -> 6514 (buffer 7168)
  const float sc = 1.f / B;      // dLoss == 1 by construction

  float lsum = 0.f;
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const float lp_t = net_fwd_sample(
        b, tid, B, training, seed, 0, nullptr, 0u, x, w1, b1, w2, b2,
        wf1,
        bf1, wf2, bf2, tgt, p1_ws, idx1_ws, m2_ws, p2_ws, idx2_ws,
        h1_ws, m3_ws, d3_ws, logp_ws, xs, w1s, p1, w2s, p2, d3, logits);
    if (tid == 0) lsum += -lp_t / B;
    __syncthreads();  // fwd LDS dead + this block's global stashes visible
    net_bwd_sample(b, 0, 1, tid, B, training, sc, wf1, wf2, tgt,
                   idx1_ws, m2_ws, idx2_ws, h1_ws, m3_ws, logp_ws,
                   glog_ws, gh1_ws, ga2_ws, ga1_ws,
                   w2s, glg, gd3, gh1, gp2, gd2p);
    __syncthreads();  // bwd LDS dead before the next sample reuses pool
  }
  if (tid == 0) loss_part[blockIdx.x] = lsum;
}

// Per-chunk partial weight gradients for all four layers in ONE launch
// (replaces 4 kernels + 8 memsets): grid.x walks tile segments
// [conv2 | fc1 | conv1 | fc2], grid.y is the batch chunk; partials land
// in part[chunk][21840] laid out exactly like the flat grad buffer
// (parameter order conv1.w,b conv2.w,b fc1.w,b fc2.w,b).
#define GW_TOTAL 21840
#define OFF_W1 0          // 250
#define OFF_B1 250        // 10
#define OFF_W2 260        // 5000
#define OFF_B2 5260       // 20
#define OFF_WF1 5280      // 16000
#define OFF_BF1 21280     // 50
#define OFF_WF2 21330     // 500
#define OFF_BF2 21830     // 10
#define T_CONV2 20        // conv2 tile columns (pair mode uses 10)
static_assert(T_CONV2 == N_C2K, "conv2 gw tiling: one tile per output "
              "channel (single mode) or per channel pair (pair mode)");
#define T_FC1 63          // ceil(16050/256)
#define T_CONV1 24        // one sub-block per conv1 output row
#define T_FC2 2           // ceil(510/256)
#define GW_TILES (T_CONV2 + T_FC1 + T_CONV1 + T_FC2)
// conv1's 24 sub-blocks write disjoint 260-float slices: sub 0 to the
// canonical [OFF_W1,OFF_B1] region, subs 1-23 to an extension past
// GW_TOTAL; the combine kernel folds the extension back in.
#define GW_ROW (GW_TOTAL + 23 * 260)

// conv2 weight-gradient fold, templated on channels-per-block so every
// accumulator index is compile-time (a runtime nk spilled q[][] to
// scratch and tripled the kernel — r2 ledger).  Thread = (ohGroup, c,
// r), 200 lanes, owns all FIVE sx weights of its (c, r) row: one
// 12-value xr window + one 8-value gr row feed 40 FMA per channel.
// Staging is double-buffered float4.
template <int NK, bool SC1 = false>
__device__ __forceinline__ void net_gw_conv2_fold(
    int k0, int tid, int b0, int b1, float* __restrict__ my,
    const float* __restrict__ p1_ws, const float* __restrict__ ga2_ws) {
  // statics INSIDE the template: the two instantiations each get an
  // allocation (the kernel carries both), but a caller-hoisted shared
  // buffer passed as pointers measured 15% SLOWER end to end (730 vs
  // 631 us gw-all at B=4096, same box) — the pointer indirection costs
  // more than the duplicated LDS at this occupancy (r2 A/B, ledger).
  __shared__ __attribute__((aligned(16))) float sp1[2][N_P1];
  __shared__ __attribute__((aligned(16))) float sg2[2][64 * NK];
  __shared__ float wacc[NK][251];
  const int ohp = tid / 50;          // 0..3 (tid < 200)
  const int cr = tid % 50;
  const int c = cr / 5, r = cr % 5;
  float q[NK][5] = {};
  float be[NK] = {}, bo[NK] = {};
  for (int i = tid; i < NK * 251; i += 256) wacc[i / 251][i % 251] = 0.f;
  // prologue: stage b0 into buffer 0 (sp1 rows and the per-sample
  // bases are 16 B aligned).  b0 < b1 guard: ceil-rounded chunk counts
  // leave EMPTY tail rows for non-divisible B (e.g. B=100, nch=16,
  // bchunk=7 -> row 15 starts at sample 105) — staging unconditionally
  // read up to ~6 KB past p1_ws/ga2_ws for those rows; the zeroed wacc
  // flush below is all an empty row needs.
  if (b0 < b1) {
    float4* d4 = reinterpret_cast<float4*>(sp1[0]);
    const float4* s4 = reinterpret_cast<const float4*>(
        p1_ws + (int64_t)b0 * N_P1);
    for (int i = tid; i < N_P1 / 4; i += 256) d4[i] = s4[i];
    float4* g4 = reinterpret_cast<float4*>(sg2[0]);
    const float4* a4 = reinterpret_cast<const float4*>(
        ga2_ws + (int64_t)b0 * N_A2 + k0 * 64);
    if (tid < 16 * NK) g4[tid] = a4[tid];
  }
  __syncthreads();
  for (int b = b0; b < b1; ++b) {
    const int cur = (b - b0) & 1;
    if (b + 1 < b1) {  // stage next sample into the other buffer
      float4* d4 = reinterpret_cast<float4*>(sp1[cur ^ 1]);
      const float4* s4 = reinterpret_cast<const float4*>(
          p1_ws + (int64_t)(b + 1) * N_P1);
      for (int i = tid; i < N_P1 / 4; i += 256) d4[i] = s4[i];
      float4* g4 = reinterpret_cast<float4*>(sg2[cur ^ 1]);
      const float4* a4 = reinterpret_cast<const float4*>(
          ga2_ws + (int64_t)(b + 1) * N_A2 + k0 * 64);
      if (tid < 16 * NK) g4[tid] = a4[tid];
    }
    if (tid < 200) {
      #pragma unroll
      for (int ohh = 0; ohh < 2; ++ohh) {
        const int oh = ohp * 2 + ohh;
        const float* xr = sp1[cur] + c * 144 + (r + oh) * 12;
        float xv[12];
        #pragma unroll
        for (int j = 0; j < 12; ++j) xv[j] = xr[j];
        #pragma unroll
        for (int kk = 0; kk < NK; ++kk) {
          const float* gr = sg2[cur] + kk * 64 + oh * 8;
          #pragma unroll
          for (int ow = 0; ow < 8; ++ow) {
            const float g = gr[ow];
            q[kk][0] += g * xv[ow];
            q[kk][1] += g * xv[ow + 1];
            q[kk][2] += g * xv[ow + 2];
            q[kk][3] += g * xv[ow + 3];
            q[kk][4] += g * xv[ow + 4];
          }
        }
      }
    } else if (tid == 200) {  // biases: two chains per channel
      #pragma unroll
      for (int kk = 0; kk < NK; ++kk) {
        const float* sg = sg2[cur] + kk * 64;
        #pragma unroll
        for (int j = 0; j < 64; j += 2) {
          be[kk] += sg[j];
          bo[kk] += sg[j + 1];
        }
      }
    }
    __syncthreads();
  }
  if (tid < 200) {
    const int e = c * 25 + r * 5;
    #pragma unroll
    for (int kk = 0; kk < NK; ++kk) {
      atomicAdd(&wacc[kk][e + 0], q[kk][0]);
      atomicAdd(&wacc[kk][e + 1], q[kk][1]);
      atomicAdd(&wacc[kk][e + 2], q[kk][2]);
      atomicAdd(&wacc[kk][e + 3], q[kk][3]);
      atomicAdd(&wacc[kk][e + 4], q[kk][4]);
    }
  } else if (tid == 200) {
    #pragma unroll
    for (int kk = 0; kk < NK; ++kk) wacc[kk][250] = be[kk] + bo[kk];
  }
  __syncthreads();
  if (tid < 250) {
    #pragma unroll
    for (int kk = 0; kk < NK; ++kk)
      gw_st<SC1>(my + OFF_W2 + (k0 + kk) * 250 + tid, wacc[kk][tid]);
  }
  if (tid == 250) {
    #pragma unroll
    for (int kk = 0; kk < NK; ++kk)
      gw_st<SC1>(my + OFF_B2 + k0 + kk, wacc[kk][250]);
  }
}

// One (tile, batch-chunk) partial weight-gradient reduction.  Tiles
// walk [conv2 | fc1 | conv1x4 | fc2]; partials land in my[GW_ROW]
// laid out like the flat grad buffer.  Shared by net_gw_partial_kernel
// and the single-launch net_step_kernel.
template <bool SC1 = false>
__device__ __forceinline__ void net_gw_tile(
    int tile, int tid, int b0, int b1, int c1_subs,
    float* __restrict__ my,
    const float* __restrict__ x,
    const float* __restrict__ p1_ws,
    const float* __restrict__ p2_ws,
    const float* __restrict__ d3_ws,
    const float* __restrict__ ga1_ws,
    const float* __restrict__ ga2_ws,
    const float* __restrict__ gh1_ws,
    const float* __restrict__ glog_ws) {
  if (tile < T_CONV2) {  // conv2: gw [20][10][5][5] + gb [20]
    // TWO block shapes, chosen by chunk size (measured, r2 ledger):
    //  * pair mode (bchunk <= 16, i.e. B <= 512): one block folds TWO
    //    output channels from one staging of the p1 plane — halves
    //    the dominant HBM re-read; tiles 10..19 idle (the grid is
    //    launch-granularity-bound here, idle blocks are free).
    //  * single mode (large B): one channel per block — smaller
    //    blocks pack the 2600-block grid without straggler tails,
    //    which beats the re-read saving end to end.
    if (b1 - b0 <= 16) {
      if (tile < N_C2K / 2)
        net_gw_conv2_fold<2, SC1>(tile * 2, tid, b0, b1, my, p1_ws,
                                  ga2_ws);
    } else {
      net_gw_conv2_fold<1, SC1>(tile, tid, b0, b1, my, p1_ws, ga2_ws);
    }
    return;
  }
  tile -= T_CONV2;
  if (tile < T_CONV1) {  // conv1: gw [10][1][5][5] + gb [10]
    // 250 outputs is too little parallelism for element-per-thread at
    // this cost (24x24 window x batch): split each output over its 24
    // output rows, LDS-atomic reduce.  The sub-block count is
    // ADAPTIVE (r2): 24 single-row subs at large B (this family was
    // the gw straggler: 3x the blocks, 281 -> 254 us at B=4096) but 8
    // three-row bands at small B, where the extra 16 extension rows
    // made the COMBINE 3x slower (8 -> 24 us) for nothing.  Sub 0
    // writes the canonical region, subs 1..c1_subs-1 the extension
    // rows; the combine folds c1_subs-1 rows.
    // (r2 note: an LDS-staged double-buffered variant was MEASURED
    // SLOWER — the grow/x re-reads are L1 hits and the staging math
    // cost more than it saved.)
    const int sub = tile;
    if (sub >= c1_subs) return;
    const int rows = 24 / c1_subs;     // 3 (band mode) or 1 (row mode)
    const int oh0 = sub * rows;
    const int n_w = 250 * rows;
    __shared__ float wacc[260];
    for (int i = tid; i < 260; i += 256) wacc[i] = 0.f;
    __syncthreads();
    for (int it = tid; it < n_w + 10 * rows; it += 256) {
      float a = 0.f;
      if (it < n_w) {
        const int e = it / rows, oh = oh0 + it % rows;
        const int k = e / 25, r = (e / 5) % 5, sx = e % 5;
        // unroll 2: each sample's rows are a cold HBM first-touch for
        // this block; interleaving two iterations overlaps the load
        // latency the serial accumulate chain was exposing (r2)
        #pragma unroll 2
        for (int b = b0; b < b1; ++b) {
          const float* grow = ga1_ws + (int64_t)b * N_A1 + k * 576 +
                              oh * 24;
          const float* xrow = x + (int64_t)b * 784 + (oh + r) * 28 + sx;
          float ae = 0.f, ao = 0.f;
          #pragma unroll
          for (int ow = 0; ow < 24; ow += 2) {
            ae += grow[ow] * xrow[ow];
            ao += grow[ow + 1] * xrow[ow + 1];
          }
          a += ae + ao;
        }
        atomicAdd(&wacc[e], a);
      } else {
        const int j = it - n_w;
        const int k = j / rows, oh = oh0 + j % rows;
        for (int b = b0; b < b1; ++b) {
          const float* grow = ga1_ws + (int64_t)b * N_A1 + k * 576 +
                              oh * 24;
          #pragma unroll 8
          for (int ow = 0; ow < 24; ++ow) a += grow[ow];
        }
        atomicAdd(&wacc[250 + k], a);
      }
    }
    __syncthreads();
    float* dst = (sub == 0) ? (my + OFF_W1) : (my + GW_TOTAL + (sub - 1) * 260);
    for (int i = tid; i < 260; i += 256) gw_st<SC1>(dst + i, wacc[i]);
    return;
  }
  tile -= T_CONV1;
  if (tile < T_FC1) {  // fc1: gw [50][320] + gb [50]
    const int i = tile * 256 + tid;
    if (i < 16050) {
      float acc = 0.f;
      if (i < 16000) {
        const int n = i / N_P2, k = i % N_P2;
        float ae = 0.f, ao = 0.f;
        int b = b0;
        for (; b + 1 < b1; b += 2) {
          ae += gh1_ws[(int64_t)b * N_H1 + n] *
                p2_ws[(int64_t)b * N_P2 + k];
          ao += gh1_ws[(int64_t)(b + 1) * N_H1 + n] *
                p2_ws[(int64_t)(b + 1) * N_P2 + k];
        }
        if (b < b1)
          ae += gh1_ws[(int64_t)b * N_H1 + n] *
                p2_ws[(int64_t)b * N_P2 + k];
        acc = ae + ao;
        gw_st<SC1>(my + OFF_WF1 + i, acc);
      } else {
        const int n = i - 16000;
        for (int b = b0; b < b1; ++b)
          acc += gh1_ws[(int64_t)b * N_H1 + n];
        gw_st<SC1>(my + OFF_BF1 + n, acc);
      }
    }
    return;
  }
  tile -= T_FC1;
  {  // fc2: gw [10][50] + gb [10]
    const int i = tile * 256 + tid;
    if (i < 510) {
      float acc = 0.f;
      if (i < 500) {
        const int n = i / N_H1, k = i % N_H1;
        for (int b = b0; b < b1; ++b)
          acc += glog_ws[(int64_t)b * N_CLS + n] *
                 d3_ws[(int64_t)b * N_H1 + k];
        gw_st<SC1>(my + OFF_WF2 + i, acc);
      } else {
        const int n = i - 500;
        for (int b = b0; b < b1; ++b)
          acc += glog_ws[(int64_t)b * N_CLS + n];
        gw_st<SC1>(my + OFF_BF2 + n, acc);
      }
    }
  }
}

__global__ void __launch_bounds__(256)
net_gw_partial_kernel(int c1_subs, const float* __restrict__ x,
                      const float* __restrict__ p1_ws,
                      const float* __restrict__ p2_ws,
                      const float* __restrict__ d3_ws,
                      const float* __restrict__ ga1_ws,
                      const float* __restrict__ ga2_ws,
                      const float* __restrict__ gh1_ws,
                      const float* __restrict__ glog_ws,
                      float* __restrict__ part,  // [max chunk][GW_ROW]
                      int B, int bchunk, int bchunk1, int bchunk2,
                      int nch, int nch1, int nch2, int tile_base) {
  // conv1/conv2 run at their own chunk counts (they are the straggler
  // families at small B); gridDim.y = max of the three, families idle
  // on the rows beyond their own count (idle blocks are
  // launch-granularity free)
  const int tile = blockIdx.x + tile_base;
  const bool c2 = tile < T_CONV2;
  const bool c1 = tile >= T_CONV2 && tile < T_CONV2 + T_CONV1;
  if ((int)blockIdx.y >= (c2 ? nch2 : c1 ? nch1 : nch)) return;
  const int bc = c2 ? bchunk2 : c1 ? bchunk1 : bchunk;
  net_gw_tile(tile, threadIdx.x, (int)blockIdx.y * bc,
              min(B, ((int)blockIdx.y + 1) * bc), c1_subs,
              part + (int64_t)blockIdx.y * GW_ROW,
              x, p1_ws, p2_ws, d3_ws, ga1_ws, ga2_ws, gh1_ws, glog_ws);
}

// combine: grads[i] = sum over chunks of part[c][i], written through
// the 8 per-parameter pointers (which may alias one flat buffer).
struct GwPtrs { float* p[8]; };
// sum flat-grad element i over the nch chunk rows (+ conv1 extension)
template <bool SC1 = false>
__device__ __forceinline__ float net_gw_combine_elem(
    int i, int nch, int nch1, int nch2, int c1_ext,
    const float* __restrict__ part) {
  // four independent accumulator chains: the single 32-deep
  // load+add chain was latency-bound (VALUBusy ~0, profiles/).
  // nch1/nch2: the conv1/conv2 families may run at their OWN chunk
  // counts (they are the gw stragglers at small B; more chunks = more
  // parallelism there without paying their fold cost on every other
  // family, which is what made a GLOBAL chunk increase lose) —
  // conv1-region elements (i < 260, extensions included) fold nch1
  // rows, conv2-region elements nch2 rows, everything else nch.
  const int n = i < 260 ? nch1
              : (i >= OFF_W2 && i < OFF_WF1) ? nch2 : nch;
  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
  int c = 0;
  for (; c + 3 < n; c += 4) {
    a0 += gw_ld<SC1>(part + (int64_t)c * GW_ROW + i);
    a1 += gw_ld<SC1>(part + (int64_t)(c + 1) * GW_ROW + i);
    a2 += gw_ld<SC1>(part + (int64_t)(c + 2) * GW_ROW + i);
    a3 += gw_ld<SC1>(part + (int64_t)(c + 3) * GW_ROW + i);
  }
  for (; c < n; ++c) a0 += gw_ld<SC1>(part + (int64_t)c * GW_ROW + i);
  if (i < 260) {  // conv1 sub-block extension rows (see GW_ROW)
    // c1_ext is 7 (band mode) or 23 (row mode): branch to fully
    // unrolled folds — a runtime-bound loop here cost the combine
    // 50% (8.1 -> 12.2 us, r2 ledger)
    if (c1_ext == 7) {
      for (int c2 = 0; c2 < nch1; ++c2) {
        const float* ext = part + (int64_t)c2 * GW_ROW + GW_TOTAL;
        a0 += gw_ld<SC1>(ext + i) + gw_ld<SC1>(ext + 4 * 260 + i);
        a1 += gw_ld<SC1>(ext + 260 + i) + gw_ld<SC1>(ext + 5 * 260 + i);
        a2 += gw_ld<SC1>(ext + 2 * 260 + i) +
              gw_ld<SC1>(ext + 6 * 260 + i);
        a3 += gw_ld<SC1>(ext + 3 * 260 + i);
      }
    } else {
      for (int c2 = 0; c2 < nch1; ++c2) {
        const float* ext = part + (int64_t)c2 * GW_ROW + GW_TOTAL;
        #pragma unroll
        for (int s = 0; s < 23; s += 4) {
          a0 += gw_ld<SC1>(ext + s * 260 + i);
          if (s + 1 < 23) a1 += gw_ld<SC1>(ext + (s + 1) * 260 + i);
          if (s + 2 < 23) a2 += gw_ld<SC1>(ext + (s + 2) * 260 + i);
          if (s + 3 < 23) a3 += gw_ld<SC1>(ext + (s + 3) * 260 + i);
        }
      }
    }
  }
  return (a0 + a1) + (a2 + a3);
}

// quad-lane variant: four consecutive lanes cooperate on one flat-grad
// element (lane q strides the chunk rows), combined with two xor
// shuffles.  Quadruples the wave count of the combine kernels — at 86
// one-element-per-thread blocks they left 2/3 of the SIMDs empty and
// ran a latency-exposed 8-deep L2 chain (profiles/).  Returns the full
// sum on EVERY lane of the quad (butterfly reduction).
__device__ __forceinline__ float net_gw_combine_elem_quad(
    int i, int q, int nch, int nch1, int nch2, int c1_ext,
    const float* __restrict__ part) {
  const int n = i < 260 ? nch1
              : (i >= OFF_W2 && i < OFF_WF1) ? nch2 : nch;
  float a0 = 0.f, a1 = 0.f;
  for (int c = q; c < n; c += 8)
    a0 += part[(int64_t)c * GW_ROW + i];
  for (int c = q + 4; c < n; c += 8)
    a1 += part[(int64_t)c * GW_ROW + i];
  if (i < 260) {  // conv1 extension rows, spread across the quad
    // fully unrolled per mode (see net_gw_combine_elem note)
    if (c1_ext == 7) {
      if (q == 0) {
        for (int c2 = 0; c2 < nch1; ++c2) {
          const float* ext = part + (int64_t)c2 * GW_ROW + GW_TOTAL;
          a0 += ext[i] + ext[260 + i] + ext[2 * 260 + i] +
                ext[3 * 260 + i];
          a1 += ext[4 * 260 + i] + ext[5 * 260 + i] + ext[6 * 260 + i];
        }
      }
    } else {
      for (int c2 = 0; c2 < nch1; ++c2) {
        const float* ext = part + (int64_t)c2 * GW_ROW + GW_TOTAL;
        #pragma unroll
        for (int s = 0; s < 23; s += 8) {
          if (s + q < 23) a0 += ext[(s + q) * 260 + i];
          if (s + q + 4 < 23) a1 += ext[(s + q + 4) * 260 + i];
        }
      }
    }
  }
  float acc = a0 + a1;
  acc += __shfl_xor(acc, 1, 4);
  acc += __shfl_xor(acc, 2, 4);
  return acc;
}

__device__ __forceinline__ int net_gw_tensor_of(int i, const int* off) {
  int t = 0;
  while (i >= off[t + 1]) ++t;
  return t;
}

// finalize the per-block loss partials written by the forward kernel
// (loss_part mode) and advance the dropout seed for the NEXT step —
// runs in block 0 of a combine kernel, replacing the step-prologue
// dispatch (~4.5 us device floor) entirely.  The seed convention is
// bump-AFTER everywhere (both the loss_part and the legacy autograd
// path): the step consumes the current seed, the combine advances it.
__device__ __forceinline__ void net_loss_finalize(
    const float* __restrict__ loss_part, float* __restrict__ loss_out,
    int nblk_fwd, unsigned long long* seed_bump) {
  if (loss_part) {
    __shared__ float red[256];
    float v = 0.f;
    for (int i = threadIdx.x; i < nblk_fwd; i += 256) v += loss_part[i];
    red[threadIdx.x] = v;
    __syncthreads();
    #pragma unroll
    for (int s2 = 128; s2 > 0; s2 >>= 1) {
      if (threadIdx.x < s2) red[threadIdx.x] += red[threadIdx.x + s2];
      __syncthreads();
    }
    if (threadIdx.x == 0) *loss_out = red[0];
  }
  if (threadIdx.x == 0 && seed_bump)
    *seed_bump += 0x9E3779B97F4A7C15ull;
}

__global__ void net_gw_combine_kernel(const float* __restrict__ part,
                                      GwPtrs g, int nch, int nch1,
                                      int nch2, int c1_ext,
                                      const float* __restrict__ loss_part,
                                      float* __restrict__ loss_out,
                                      int nblk_fwd,
                                      unsigned long long* seed_bump) {
  const int off[9] = {OFF_W1, OFF_B1, OFF_W2, OFF_B2, OFF_WF1, OFF_BF1,
                      OFF_WF2, OFF_BF2, GW_TOTAL};
  for (int64_t t4 = blockIdx.x * blockDim.x + threadIdx.x;
       t4 < (int64_t)GW_TOTAL * 4;
       t4 += (int64_t)gridDim.x * blockDim.x) {
    const int i = (int)(t4 >> 2), q = (int)(t4 & 3);
    const float acc = net_gw_combine_elem_quad(i, q, nch, nch1, nch2,
                                               c1_ext, part);
    if (q == 0) {
      const int t = net_gw_tensor_of(i, off);
      g.p[t][i - off[t]] = acc;
    }
  }
  if ((loss_part || seed_bump) && blockIdx.x == 0)
    net_loss_finalize(loss_part, loss_out, nblk_fwd, seed_bump);
}

// combine + SGD in one dispatch (single-GPU training: there is no
// gradient all-reduce between combine and the optimizer step, so the
// ~4.5 us dispatch floor of the separate sgd_step_kernel is pure
// overhead).  Writes the grad (so .grad stays inspectable), updates
// the momentum buffer and the parameter exactly like sgd_step_kernel.
__global__ void net_gw_combine_sgd_kernel(const float* __restrict__ part,
                                          GwPtrs g, GwPtrs prm,
                                          GwPtrs buf, int nch, int nch1,
                                          int nch2, int c1_ext,
                                          float lr,
                                          float mu,
                                          const float* __restrict__ loss_part,
                                          float* __restrict__ loss_out,
                                          int nblk_fwd,
                                          unsigned long long* seed_bump) {
  const int off[9] = {OFF_W1, OFF_B1, OFF_W2, OFF_B2, OFF_WF1, OFF_BF1,
                      OFF_WF2, OFF_BF2, GW_TOTAL};
  for (int64_t t4 = blockIdx.x * blockDim.x + threadIdx.x;
       t4 < (int64_t)GW_TOTAL * 4;
       t4 += (int64_t)gridDim.x * blockDim.x) {
    const int i = (int)(t4 >> 2), q = (int)(t4 & 3);
    const float acc = net_gw_combine_elem_quad(i, q, nch, nch1, nch2,
                                               c1_ext, part);
    if (q == 0) {
      const int t = net_gw_tensor_of(i, off);
      const int64_t j = i - off[t];
      g.p[t][j] = acc;
      float v = acc;
      if (buf.p[t]) {
        v = mu * buf.p[t][j] + acc;
        buf.p[t][j] = v;
      }
      prm.p[t][j] -= lr * v;
    }
  }
  if ((loss_part || seed_bump) && blockIdx.x == 0)
    net_loss_finalize(loss_part, loss_out, nblk_fwd, seed_bump);
}

// ---------------------------------------------------------------------------
// In-launch gw fold (r2): the combine dispatch is pure latency (~10 us at
// B=128 — 86 tiny blocks, an L2-chain fold of 1.4 MB of partials), so this
// variant of the partial kernel deletes it: the LAST-ARRIVING block of each
// tile column folds the region its column wrote (and, DO_SGD, applies the
// momentum update), using the split-K last-arriver recipe from the CDNA
// guide (G16): every block drains vmcnt + __syncthreads, lane 0 issues an
// agent-scope RELEASE fence, restates the vmcnt wait, THEN takes a relaxed
// agent-scope ticket; the block that draws target-1 acquires (agent) and
// reads every row with plain loads.  Fold regions are per COLUMN (what the
// column's own blocks wrote), so counter==target implies the region's rows
// are all visible:
//   conv2 column k  -> channel k (pair mode: column k<10 -> channels 2k,
//                      2k+1; columns 10..19 wrote nothing and own nothing)
//   conv1 family    -> ONE counter over all T_CONV1*nch blocks (its
//                      sub-blocks share the 260 outputs via extension rows,
//                      so no single column's completion suffices)
//   fc1/fc2 column j -> its 256-element flat slice
// Each fold runs as soon as ITS column's blocks are done — overlapped with
// the other families still computing — so the only exposed cost is the
// straggler family's own (tiny) fold.
__device__ __forceinline__ void net_gw_fold_range(
    int i0, int i1, int nch, int c1_ext, const float* __restrict__ part,
    const GwPtrs& g, const GwPtrs& prm, const GwPtrs& buf, float lr,
    float mu, bool do_sgd) {
  const int off[9] = {OFF_W1, OFF_B1, OFF_W2, OFF_B2, OFF_WF1, OFF_BF1,
                      OFF_WF2, OFF_BF2, GW_TOTAL};
  for (int i = i0 + (int)threadIdx.x; i < i1; i += 256) {
    const float acc = net_gw_combine_elem<true>(i, nch, nch, nch,
                                                c1_ext, part);
    const int t = net_gw_tensor_of(i, off);
    const int64_t j = i - off[t];
    g.p[t][j] = acc;
    if (do_sgd) {
      float v = acc;
      if (buf.p[t]) {
        v = mu * buf.p[t][j] + acc;
        buf.p[t][j] = v;
      }
      prm.p[t][j] -= lr * v;
    }
  }
}

template <bool DO_SGD>
__global__ void __launch_bounds__(256)
net_gw_partial_fold_kernel(int c1_subs, const float* __restrict__ x,
                           const float* __restrict__ p1_ws,
                           const float* __restrict__ p2_ws,
                           const float* __restrict__ d3_ws,
                           const float* __restrict__ ga1_ws,
                           const float* __restrict__ ga2_ws,
                           const float* __restrict__ gh1_ws,
                           const float* __restrict__ glog_ws,
                           float* __restrict__ part, int B, int bchunk,
                           GwPtrs g, GwPtrs prm, GwPtrs buf, float lr,
                           float mu, const float* __restrict__ loss_part,
                           float* __restrict__ loss_out, int nblk_fwd,
                           unsigned long long* seed_bump,
                           unsigned int* __restrict__ cnt) {
  const int b0 = blockIdx.y * bchunk;
  net_gw_tile<true>(blockIdx.x, threadIdx.x, b0, min(B, b0 + bchunk),
                    c1_subs, part + (int64_t)blockIdx.y * GW_ROW,
                    x, p1_ws, p2_ws, d3_ws, ga1_ws, ga2_ws, gh1_ws,
                    glog_ws);
  // ---- last-arriver ticket, FENCE-FREE: the tile wrote its partials
  // with sc1 stores (already at the coherence point once vmcnt
  // retires), so the ticket needs only the vmcnt drain before it and
  // the reducer reads the rows back with sc1 loads.  No buffer_wbl2 /
  // buffer_inv anywhere — the concurrent tile blocks keep their
  // L1/L2-resident activations (the fenced variant cost 23-28%).
  const int nch = (int)gridDim.y;
  const int xcol = (int)blockIdx.x;
  const bool conv1_fam = (xcol >= T_CONV2 && xcol < T_CONV2 + T_CONV1);
  const int ci = conv1_fam ? T_CONV2 : xcol;
  const unsigned target =
      conv1_fam ? (unsigned)(T_CONV1 * nch) : (unsigned)nch;
  __shared__ unsigned s_tk;
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0)
    s_tk = __hip_atomic_fetch_add(&cnt[ci], 1u, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
  __syncthreads();
  if (s_tk != target - 1) return;  // not the last arriver of this region
  if (threadIdx.x == 0)
    __hip_atomic_store(&cnt[ci], 0u, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_AGENT);  // re-arm for next step
  __syncthreads();
  const int c1_ext = c1_subs - 1;
  if (xcol < T_CONV2) {  // conv2: this column's channel(s)
    int k0, nk;
    if (bchunk <= 16) {  // pair mode (mirrors net_gw_tile's split)
      if (xcol >= N_C2K / 2) return;  // idle pair columns wrote nothing
      k0 = xcol * 2;
      nk = 2;
    } else {
      k0 = xcol;
      nk = 1;
    }
    net_gw_fold_range(OFF_W2 + k0 * 250, OFF_W2 + (k0 + nk) * 250, nch,
                      c1_ext, part, g, prm, buf, lr, mu, DO_SGD);
    net_gw_fold_range(OFF_B2 + k0, OFF_B2 + k0 + nk, nch, c1_ext, part,
                      g, prm, buf, lr, mu, DO_SGD);
    return;
  }
  if (conv1_fam) {  // conv1: canonical 260 + extension rows (c1_ext)
    net_gw_fold_range(0, 260, nch, c1_ext, part, g, prm, buf, lr, mu,
                      DO_SGD);
    return;
  }
  if (xcol < T_CONV2 + T_CONV1 + T_FC1) {  // fc1 column slice
    const int i0 = OFF_WF1 + (xcol - (T_CONV2 + T_CONV1)) * 256;
    net_gw_fold_range(i0, min(OFF_WF1 + 16050, i0 + 256), nch, c1_ext,
                      part, g, prm, buf, lr, mu, DO_SGD);
    return;
  }
  // fc2 column slice; the last column also finalizes the loss partials
  // and advances the dropout seed (wave 0 only — no extra LDS).
  const int i0 = OFF_WF2 + (xcol - (T_CONV2 + T_CONV1 + T_FC1)) * 256;
  net_gw_fold_range(i0, min(OFF_WF2 + 510, i0 + 256), nch, c1_ext, part,
                    g, prm, buf, lr, mu, DO_SGD);
  if (xcol == GW_TILES - 1 && threadIdx.x < 64) {
    if (loss_part) {
      float v = 0.f;
      for (int i = threadIdx.x; i < nblk_fwd; i += 64) v += loss_part[i];
      #pragma unroll
      for (int s = 32; s > 0; s >>= 1) v += __shfl_down(v, s, 64);
      if (threadIdx.x == 0 && loss_out) *loss_out = v;
    }
    if (threadIdx.x == 0 && seed_bump)
      *seed_bump += 0x9E3779B97F4A7C15ull;
  }
}

// ===========================================================================
// Single-launch training step (cooperative kernel).
//
// The 6-dispatch step (prologue, fwd, bwd, gw-partial, combine, sgd)
// pays a ~4.5 us device dispatch floor per kernel at the reference's
// batch sizes (rocprof, profiles/).  This kernel runs the WHOLE step —
// forward, data backward, weight-gradient reduction, combine and the
// SGD+momentum update — in ONE hipLaunchCooperativeKernel dispatch,
// with cooperative-groups grid barriers between phases.  The per-phase
// work is the same shared __device__ code as the modular kernels
// (net_fwd_sample / net_bwd_sample / net_gw_tile / net_gw_combine_elem)
// so numerics are identical by construction.
//
// The loss is accumulated per-workgroup (loss_part, no atomics and no
// pre-zeroing) and summed into loss_out by workgroup 0 in the last
// phase, which also advances the device dropout seed for the NEXT step
// (replacing the step-prologue kernel; hipGraph-replay safe).
//
// When do_sgd == 0 the kernel stops after writing grads (the DP path:
// host runs the flat-gradient all-reduce, then sgd_step_kernel).
// ===========================================================================
namespace cg = cooperative_groups;

__global__ void __launch_bounds__(256)
net_step_kernel(
    const float* __restrict__ x,
    const int64_t* __restrict__ tgt,
    const float* __restrict__ gl,
    float* __restrict__ p1_ws, uint8_t* __restrict__ idx1_ws,
    uint8_t* __restrict__ m2_ws, float* __restrict__ p2_ws,
    uint8_t* __restrict__ idx2_ws, float* __restrict__ h1_ws,
    uint8_t* __restrict__ m3_ws, float* __restrict__ d3_ws,
    float* __restrict__ logp_ws, float* __restrict__ glog_ws,
    float* __restrict__ gh1_ws, float* __restrict__ ga2_ws,
    float* __restrict__ ga1_ws, float* __restrict__ part,
    float* __restrict__ loss_part,    // [gridDim.x]
    float* __restrict__ loss_out,     // scalar
    unsigned long long* __restrict__ seed_p,
    GwPtrs prm, GwPtrs grd, GwPtrs buf,
    float lr, float mu, int do_sgd,
    int B, int training, int split, int bchunk, int nch) {
  // LDS union: bwd carve (11,872 floats) reused by the fwd carve
  // (7,884) and the final-phase reduction scratch; phases are
  // separated by grid barriers.
  __shared__ __attribute__((aligned(16))) float smem[12192];
  float* xs = smem;             // 784
  float* w1s = xs + 784;        // 260
  float* p1 = w1s + 260;        // 1440
  float* w2s = p1 + 1440;       // 5020 (fwd: w2+b2)
  float* p2 = w2s + 5020;       // 320
  float* d3 = p2 + 320;         // 50
  float* logits = d3 + 50;      // 10
  float* b_w2s = smem;          // 5000 (bwd: w2 only)
  float* b_glg = smem + 5008;   // 10
  float* b_gd3 = smem + 5024;   // 50
  float* b_gh1 = smem + 5088;   // 50
  float* b_gp2 = smem + 5152;   // 320
  float* b_gd2p = smem + 5472;  // 6080 (20 x 16 x 19) -> 11552
  const int tid = threadIdx.x;
  const int wg = blockIdx.x, nblk = gridDim.x;
  cg::grid_group grid = cg::this_grid();

  const float* w1 = prm.p[0]; const float* b1 = prm.p[1];
  const float* w2 = prm.p[2]; const float* b2 = prm.p[3];
  const float* wf1 = prm.p[4]; const float* bf1 = prm.p[5];
  const float* wf2 = prm.p[6]; const float* bf2 = prm.p[7];
  const uint64_t seed = seed_p[0];

  // ---- phase 1: forward ------------------------------------------------
  float lsum = 0.f;
  for (int b = wg; b < B; b += nblk) {
    const float lp_t = net_fwd_sample(
        b, tid, B, training, seed, 0, nullptr, 0u, x, w1, b1, w2, b2,
        wf1,
        bf1, wf2, bf2, tgt, p1_ws, idx1_ws, m2_ws, p2_ws, idx2_ws,
        h1_ws, m3_ws, d3_ws, logp_ws, xs, w1s, p1, w2s, p2, d3, logits);
    if (tid == 0) lsum += -lp_t / B;
  }
  if (tid == 0) loss_part[wg] = lsum;
  grid.sync();

  // ---- phase 2: data backward -----------------------------------------
  for (int i = tid; i < N_C2K * 250; i += 256) b_w2s[i] = w2[i];
  const float sc = gl[0] / B;
  for (int bb = wg; bb < B * split; bb += nblk) {
    net_bwd_sample(bb / split, bb % split, split, tid, B, training, sc,
                   wf1, wf2, tgt, idx1_ws, m2_ws, idx2_ws, h1_ws, m3_ws,
                   logp_ws, glog_ws, gh1_ws, ga2_ws, ga1_ws,
                   b_w2s, b_glg, b_gd3, b_gh1, b_gp2, b_gd2p);
  }
  grid.sync();

  // ---- phase 3: partial weight gradients ------------------------------
  for (int t = wg; t < GW_TILES * nch; t += nblk) {
    const int tile = t % GW_TILES, ch = t / GW_TILES;
    const int b0 = ch * bchunk;
    net_gw_tile(tile, tid, b0, min(B, b0 + bchunk), 8,
                part + (int64_t)ch * GW_ROW, x, p1_ws, p2_ws, d3_ws,
                ga1_ws, ga2_ws, gh1_ws, glog_ws);
    __syncthreads();
  }
  grid.sync();

  // ---- phase 4: combine (+ SGD), loss finalize, seed bump --------------
  const int off[9] = {OFF_W1, OFF_B1, OFF_W2, OFF_B2, OFF_WF1, OFF_BF1,
                      OFF_WF2, OFF_BF2, GW_TOTAL};
  for (int i = wg * 256 + tid; i < GW_TOTAL; i += nblk * 256) {
    const float acc = net_gw_combine_elem(i, nch, nch, nch, 7, part);
    const int t = net_gw_tensor_of(i, off);
    const int64_t j = i - off[t];
    grd.p[t][j] = acc;
    if (do_sgd) {
      float v = acc;
      if (buf.p[t]) {
        v = mu * buf.p[t][j] + acc;
        buf.p[t][j] = v;
      }
      prm.p[t][j] -= lr * v;
    }
  }
  if (wg == 0) {
    float v = (tid < nblk) ? loss_part[tid] : 0.f;
    if (tid + 256 < nblk) v += loss_part[tid + 256];  // nblk <= 512
    smem[tid] = v;
    __syncthreads();
    for (int s = 128; s > 0; s >>= 1) {
      if (tid < s) smem[tid] += smem[tid + s];
      __syncthreads();
    }
    if (tid == 0) {
      *loss_out = smem[0];
      if (training) *seed_p = seed + 0x9E3779B97F4A7C15ull;
    }
  }
}

// ===========================================================================
// host launchers + pybind
// ===========================================================================
namespace {

constexpr int BLK = 256;

void conv2d_fwd(uintptr_t x, uintptr_t w, uintptr_t bias, uintptr_t out,
                int B, int C, int H, int W, int K, int R, int S_,
                uintptr_t stream) {
  const int64_t need = (int64_t)K * C * R * S_ * sizeof(float);
  const int staged = need <= 64 * 1024 ? 1 : 0;
  const int OH = H - R + 1, OW = W - S_ + 1;
  const int64_t n = (int64_t)B * K * OH * OW;
  hipLaunchKernelGGL(conv2d_fwd_kernel, dim3(grid_for(n, BLK)), dim3(BLK),
                     staged ? (int)need : 0, S(stream), (const float*)x,
                     (const float*)w, (const float*)bias, (float*)out, B,
                     C, H, W, K, R, S_, staged);
}

void conv2d_bwd(uintptr_t x, uintptr_t w, uintptr_t gy, uintptr_t gx,
                uintptr_t gw, uintptr_t gb, int B, int C, int H, int W,
                int K, int R, int S_, uintptr_t stream) {
  const int OH = H - R + 1, OW = W - S_ + 1;
  {
    const int64_t need = (int64_t)K * C * R * S_ * sizeof(float);
    const int staged = need <= 64 * 1024 ? 1 : 0;
    const int64_t n = (int64_t)B * C * H * W;
    hipLaunchKernelGGL(conv2d_bwd_x_kernel, dim3(grid_for(n, BLK)),
                       dim3(BLK), staged ? (int)need : 0, S(stream),
                       (const float*)gy, (const float*)w, (float*)gx, B,
                       C, H, W, K, R, S_, staged);
  }
  {
    const int wn = K * C * R * S_;
    // pick the batch chunk so the grid lands near ~512 blocks
    int bchunk = B;
    const int wtiles = (wn + BLK - 1) / BLK;
    while (bchunk > 1 && wtiles * ((B + bchunk - 1) / bchunk) < 512)
      bchunk = (bchunk + 1) / 2;
    const int nchunks = (B + bchunk - 1) / bchunk;
    if (nchunks > 1) {
      HIP_CHECK(hipMemsetAsync((void*)gw, 0, (size_t)wn * sizeof(float),
                               S(stream)));
      if (gb)
        HIP_CHECK(hipMemsetAsync((void*)gb, 0, K * sizeof(float),
                                 S(stream)));
    }
    hipLaunchKernelGGL(conv2d_bwd_w_kernel, dim3(wtiles, nchunks),
                       dim3(BLK), 0, S(stream), (const float*)x,
                       (const float*)gy, (float*)gw, (float*)gb, B, C, H,
                       W, K, R, S_, bchunk);
  }
}

void maxpool2d_relu_fwd(uintptr_t x, uintptr_t out, uintptr_t idx, int B,
                        int C, int H, int W, uintptr_t stream) {
  const int64_t planes = (int64_t)B * C;
  const int64_t n = planes * (H / 2) * (W / 2);
  hipLaunchKernelGGL(maxpool2d_relu_fwd_kernel, dim3(grid_for(n, BLK)),
                     dim3(BLK), 0, S(stream), (const float*)x, (float*)out,
                     (int*)idx, planes, H, W);
}

void maxpool2d_relu_bwd(uintptr_t gy, uintptr_t idx, uintptr_t gx, int B,
                        int C, int H, int W, uintptr_t stream) {
  const int64_t planes = (int64_t)B * C;
  const int64_t n = planes * (H / 2) * (W / 2);
  hipLaunchKernelGGL(maxpool2d_relu_bwd_kernel, dim3(grid_for(n, BLK)),
                     dim3(BLK), 0, S(stream), (const float*)gy,
                     (const int*)idx, (float*)gx, planes, H, W);
}

void relu_fwd(uintptr_t x, uintptr_t out, int64_t n, uintptr_t stream) {
  hipLaunchKernelGGL(relu_fwd_kernel, dim3(grid_for(n, BLK, 4)), dim3(BLK),
                     0, S(stream), (const float*)x, (float*)out, n);
}

void relu_bwd(uintptr_t gy, uintptr_t out, uintptr_t gx, int64_t n,
              uintptr_t stream) {
  hipLaunchKernelGGL(relu_bwd_kernel, dim3(grid_for(n, BLK)), dim3(BLK), 0,
                     S(stream), (const float*)gy, (const float*)out,
                     (float*)gx, n);
}

void dropout_fwd(uintptr_t x, uintptr_t out, uintptr_t mask, int64_t n,
                 double p, uintptr_t seed_dev, uintptr_t stream) {
  const float scale = 1.f / (1.f - (float)p);
  hipLaunchKernelGGL(bump_seed_kernel, dim3(1), dim3(64), 0, S(stream),
                     (unsigned long long*)seed_dev);
  hipLaunchKernelGGL(dropout_fwd_kernel, dim3(grid_for(n, BLK)), dim3(BLK),
                     0, S(stream), (const float*)x, (float*)out,
                     (uint8_t*)mask, n, (float)p, scale,
                     (const unsigned long long*)seed_dev);
}

void dropout_bwd(uintptr_t gy, uintptr_t mask, uintptr_t gx, int64_t n,
                 double scale, uintptr_t stream) {
  hipLaunchKernelGGL(dropout_bwd_kernel, dim3(grid_for(n, BLK)), dim3(BLK),
                     0, S(stream), (const float*)gy, (const uint8_t*)mask,
                     (float*)gx, n, (float)scale);
}

void dropout2d_fwd(uintptr_t x, uintptr_t out, uintptr_t mask,
                   int64_t planes, int64_t hw, double p,
                   uintptr_t seed_dev, uintptr_t stream) {
  const float scale = 1.f / (1.f - (float)p);
  hipLaunchKernelGGL(bump_seed_kernel, dim3(1), dim3(64), 0, S(stream),
                     (unsigned long long*)seed_dev);
  hipLaunchKernelGGL(dropout2d_fwd_kernel,
                     dim3(grid_for(planes * hw, BLK)), dim3(BLK), 0,
                     S(stream), (const float*)x, (float*)out,
                     (uint8_t*)mask, planes, hw, (float)p, scale,
                     (const unsigned long long*)seed_dev);
}

void dropout2d_bwd(uintptr_t gy, uintptr_t mask, uintptr_t gx,
                   int64_t planes, int64_t hw, double scale,
                   uintptr_t stream) {
  hipLaunchKernelGGL(dropout2d_bwd_kernel,
                     dim3(grid_for(planes * hw, BLK)), dim3(BLK), 0,
                     S(stream), (const float*)gy, (const uint8_t*)mask,
                     (float*)gx, planes, hw, (float)scale);
}

void linear_fwd(uintptr_t x, uintptr_t w, uintptr_t bias, uintptr_t out,
                int B, int K, int N, bool fuse_relu, uintptr_t stream) {
  const int64_t need = (int64_t)N * K * sizeof(float);
  const int staged = need <= 64 * 1024 ? 1 : 0;  // LDS budget per WG
  hipLaunchKernelGGL(linear_fwd_kernel,
                     dim3(grid_for((int64_t)B * N, BLK)), dim3(BLK),
                     staged ? (int)need : 0,
                     S(stream), (const float*)x, (const float*)w,
                     (const float*)bias, (float*)out, B, K, N,
                     fuse_relu ? 1 : 0, staged);
}

void linear_bwd(uintptr_t x, uintptr_t w, uintptr_t gy, uintptr_t out,
                uintptr_t gx, uintptr_t gw, uintptr_t gb, int B, int K,
                int N, uintptr_t stream) {
  {
    const int64_t need = (int64_t)N * K * sizeof(float);
    const int staged = need <= 64 * 1024 ? 1 : 0;
    hipLaunchKernelGGL(linear_bwd_x_kernel,
                       dim3(grid_for((int64_t)B * K, BLK)), dim3(BLK),
                       staged ? (int)need : 0,
                       S(stream), (const float*)gy, (const float*)out,
                       (const float*)w, (float*)gx, B, K, N, staged);
  }
  {
    const int wn = N * K;
    const int wtiles = (wn + BLK - 1) / BLK;
    int bchunk = B;
    while (bchunk > 1 && wtiles * ((B + bchunk - 1) / bchunk) < 512)
      bchunk = (bchunk + 1) / 2;
    const int nchunks = (B + bchunk - 1) / bchunk;
    if (nchunks > 1) {
      HIP_CHECK(hipMemsetAsync((void*)gw, 0, (size_t)wn * sizeof(float),
                               S(stream)));
      if (gb)
        HIP_CHECK(hipMemsetAsync((void*)gb, 0, N * sizeof(float),
                                 S(stream)));
    }
    hipLaunchKernelGGL(linear_bwd_w_kernel, dim3(wtiles, nchunks),
                       dim3(BLK), 0, S(stream), (const float*)x,
                       (const float*)gy, (const float*)out, (float*)gw,
                       (float*)gb, B, K, N, bchunk);
  }
}

void log_softmax_fwd(uintptr_t x, uintptr_t out, int B, int N,
                     uintptr_t stream) {
  hipLaunchKernelGGL(log_softmax_fwd_kernel, dim3(grid_for(B, BLK)),
                     dim3(BLK), 0, S(stream), (const float*)x, (float*)out,
                     B, N);
}

void log_softmax_bwd(uintptr_t gy, uintptr_t out, uintptr_t gx, int B,
                     int N, uintptr_t stream) {
  hipLaunchKernelGGL(log_softmax_bwd_kernel, dim3(grid_for(B, BLK)),
                     dim3(BLK), 0, S(stream), (const float*)gy,
                     (const float*)out, (float*)gx, B, N);
}

void nll_loss_fwd(uintptr_t logp, uintptr_t target, uintptr_t loss, int B,
                  int N, uintptr_t stream) {
  HIP_CHECK(hipMemsetAsync((void*)loss, 0, sizeof(float), S(stream)));
  hipLaunchKernelGGL(nll_loss_fwd_kernel, dim3(grid_for(B, BLK)), dim3(BLK),
                     0, S(stream), (const float*)logp,
                     (const int64_t*)target, (float*)loss, B, N);
}

void nll_loss_bwd(uintptr_t target, uintptr_t gx, uintptr_t gloss, int B,
                  int N, uintptr_t stream) {
  hipLaunchKernelGGL(nll_loss_bwd_kernel, dim3(grid_for(B, BLK)), dim3(BLK),
                     0, S(stream), (const int64_t*)target, (float*)gx,
                     (const float*)gloss, B, N);
}

void log_softmax_nll_fwd(uintptr_t x, uintptr_t tgt, uintptr_t logp,
                         uintptr_t loss, int B, int N, uintptr_t stream) {
  HIP_CHECK(hipMemsetAsync((void*)loss, 0, sizeof(float), S(stream)));
  hipLaunchKernelGGL(log_softmax_nll_fwd_kernel, dim3(grid_for(B, BLK)),
                     dim3(BLK), 0, S(stream), (const float*)x,
                     (const int64_t*)tgt, (float*)logp, (float*)loss, B, N);
}

void log_softmax_nll_bwd(uintptr_t logp, uintptr_t tgt, uintptr_t gx,
                         uintptr_t gloss, int B, int N, uintptr_t stream) {
  hipLaunchKernelGGL(log_softmax_nll_bwd_kernel, dim3(grid_for(B, BLK)),
                     dim3(BLK), 0, S(stream), (const float*)logp,
                     (const int64_t*)tgt, (float*)gx, (const float*)gloss,
                     B, N);
}

void sgd_step(const std::vector<uintptr_t>& ps,
              const std::vector<uintptr_t>& gs,
              const std::vector<uintptr_t>& bufs,
              const std::vector<int64_t>& numels, double lr, double mu,
              bool zero_grad, uintptr_t stream) {
  size_t i = 0;
  while (i < ps.size()) {
    MtArgs a{};
    a.count = 0;
    a.total = 0;
    while (i < ps.size() && a.count < MT_MAX) {
      const int c = a.count;
      a.p[c] = (float*)ps[i];
      a.g[c] = (float*)gs[i];
      a.buf[c] = (float*)bufs[i];
      a.numel[c] = numels[i];
      a.offset[c] = a.total;
      a.total += numels[i];
      ++a.count;
      ++i;
    }
    hipLaunchKernelGGL(sgd_step_kernel, dim3(grid_for(a.total, BLK)),
                       dim3(BLK), 0, S(stream), a, (float)lr, (float)mu,
                       zero_grad ? 1 : 0);
  }
}


// conv1 sub-block count: 8 three-row bands at small chunks (a cheap
// combine fold), 24 single-row subs at large chunks (3x the blocks
// where conv1 is the gw straggler) — r2 ledger.
static inline int gw_c1_subs(int bchunk) { return bchunk <= 16 ? 8 : 24; }

// gw chunk count: default 32 (or B when smaller); DTP_GW_NCH overrides.
static int gw_nch(int B) {
  static int env_nch = -2;
  if (env_nch == -2) {
    const char* e = std::getenv("DTP_GW_NCH");
    env_nch = e ? std::atoi(e) : -1;
  }
  int nch;
  if (env_nch > 0) {
    nch = env_nch;
  } else if (B <= 192) {
    nch = 16;   // B=128: 1.67M vs 1.61M at 32 (r2 sweep, post-rework)
  } else if (B <= 768) {
    nch = 24;   // B=512: 3.31M vs 3.06M at 32
  } else {
    nch = 32;   // B>=1024: 32 still best (B=4096: 4.76M vs 4.61M at 24)
  }
  if (nch > 32) nch = 32;  // the part workspace holds 32 rows (_ws)
  if (nch > B) nch = B;
  return nch;
}

// in-launch gw fold (last-arriver epilogue in the partial kernel, no
// combine dispatch).  DTP_GW_FOLD=1 opts in; the default stays the
// two-kernel path — MEASURED NEGATIVE on both step shapes (ledger):
// the combine's dispatch latency is hidden by the stream pipeline
// (kernels enqueue back-to-back), so only its ~5 us execution is
// exposed, while the epilogue's per-block sc1-drain tail costs ~4-5 us
// across the critical path at B=128 (-5%); parity at B>=512.
static bool gw_fold_on() {
  static int v = -1;
  if (v < 0) {
    const char* e = std::getenv("DTP_GW_FOLD");
    v = (e && std::atoi(e) != 0) ? 1 : 0;
  }
  return v == 1;
}

// per-device ticket counters for the fold epilogue (GW_TILES u32,
// zeroed once; each reducer re-arms its own counter every step)
static unsigned int* gw_cnt_buf(hipStream_t s) {
  static unsigned int* bufs[64] = {};
  int dev = 0;
  HIP_CHECK(hipGetDevice(&dev));
  if (dev < 0 || dev >= 64) throw std::runtime_error("device index > 64");
  if (!bufs[dev]) {
    HIP_CHECK(hipMalloc(&bufs[dev], GW_TILES * sizeof(unsigned int)));
    HIP_CHECK(hipMemsetAsync(bufs[dev], 0, GW_TILES * sizeof(unsigned int),
                             s));
  }
  return bufs[dev];
}

// fwd sibling split: 2 workgroups per sample when one per sample
// cannot fill the 256 CUs.  DTP_FWD_SPLIT=1 forces off, =2 forces on.
static int fwd_split(int B) {
  static int env_v = -2;
  if (env_v == -2) {
    const char* e = std::getenv("DTP_FWD_SPLIT");
    env_v = e ? std::atoi(e) : -1;
  }
  if (B * 2 > 4096) return 1;  // loss_part/flags hold 4096 entries
  if (env_v == 1) return 1;
  if (env_v == 2) return 2;
  // 3-rep A/B (profiles/sweeps/fsplit_reps.log): +1.5% at B=64, +0.6% at
  // B=128, -2.5% at B=192 (the pair grid exceeds the CU count there
  // and the exchange round trip is pure overhead) — so the split is
  // a small-batch lever only.
  return B <= 128 ? 2 : 1;
}

// fwd grid size (the combine kernels fold loss_part over exactly this
// many per-block partials)
static int fwd_grid(int B) {
  const int sp = fwd_split(B);
  return sp == 2 ? 2 * B : grid_for(B, 1);
}

// monotone per-process token for the split-fwd handshake: each launch
// publishes THIS value, so a stale write from a previous step's late
// sibling can never satisfy a later step's wait.  (Never returns 0,
// the flags' initial state.)
static unsigned int fwd_token() {
  static unsigned int t = 0;
  if (++t == 0) ++t;
  return t;
}

// per-device per-sample handshake flags for the split fwd (zeroed
// once; published tokens are step-unique, no re-arm needed)
static unsigned int* fwd_flags_buf(hipStream_t s) {
  static unsigned int* bufs[64] = {};
  int dev = 0;
  HIP_CHECK(hipGetDevice(&dev));
  if (dev < 0 || dev >= 64) throw std::runtime_error("device index > 64");
  if (!bufs[dev]) {
    HIP_CHECK(hipMalloc(&bufs[dev], 4096 * sizeof(unsigned int)));
    HIP_CHECK(hipMemsetAsync(bufs[dev], 0, 4096 * sizeof(unsigned int),
                             s));
  }
  return bufs[dev];
}

// conv2-family gw chunk count (DTP_GW_NCH2 overrides).  The conv2
// pair-mode columns are the gw straggler at small B (20.2 us isolated
// of the 29.5 us bundle at B=128, kernel_micro); running ONLY conv2 at
// a higher chunk count buys its parallelism without the conv1
// extension-fold cost that made a GLOBAL chunk increase lose.
// conv1-family gw chunk count (DTP_GW_NCH1 overrides); 16.6 us
// isolated at B=128 — the second straggler after conv2.  Its combine
// cost scales with nch1 * c1_ext (extension rows), which is why the
// global sweep rejected 32; family-local it can still pay.
static int gw_nch1(int B, int nch) {
  static int env_nch = -2;
  if (env_nch == -2) {
    const char* e = std::getenv("DTP_GW_NCH1");
    env_nch = e ? std::atoi(e) : -1;
  }
  int nch1 = env_nch > 0 ? env_nch : nch;
  if (nch1 > 32) nch1 = 32;
  if (nch1 > B) nch1 = B;
  return nch1;
}

static int gw_nch2(int B, int nch) {
  static int env_nch = -2;
  if (env_nch == -2) {
    const char* e = std::getenv("DTP_GW_NCH2");
    env_nch = e ? std::atoi(e) : -1;
  }
  // sweep (profiles/sweeps/nch2_sweep.log): B=128 nch2=24 74.6us vs 76.2 at
  // the uniform 16 (+2.1%); 32 within noise of 24; B=512 nch2=32 LOSES
  // (combine cost) — so only the small-batch band deviates from nch.
  int nch2 = env_nch > 0 ? env_nch : (B <= 192 ? 24 : nch);
  if (nch2 > 32) nch2 = 32;  // part workspace holds 32 rows (_ws)
  if (nch2 > B) nch2 = B;
  return nch2;
}

void net_fused_fwd(uintptr_t x, uintptr_t w1, uintptr_t b1, uintptr_t w2,
                   uintptr_t b2, uintptr_t wf1, uintptr_t bf1,
                   uintptr_t wf2, uintptr_t bf2, uintptr_t tgt,
                   uintptr_t p1_ws, uintptr_t idx1_ws, uintptr_t m2_ws,
                   uintptr_t p2_ws, uintptr_t idx2_ws, uintptr_t h1_ws,
                   uintptr_t m3_ws, uintptr_t d3_ws, uintptr_t logp_ws,
                   uintptr_t loss, uintptr_t loss_part,
                   uintptr_t seed_dev, int B,
                   bool training, uintptr_t stream) {
  // legacy mode (loss_part == 0): prologue dispatch zeroes the loss
  // scalar, fwd accumulates the loss atomically.  loss_part mode: no
  // prologue — per-block partials, finalized by the combine kernel.
  // Both modes bump the dropout seed AFTER the step, in the combine
  // kernel of net_fused_bwd (one convention, so the autograd path and
  // the fused-step path can interleave without mask reuse).
  if (!loss_part)
    hipLaunchKernelGGL(step_prologue_kernel, dim3(1), dim3(64), 0,
                       S(stream), (unsigned long long*)nullptr,
                       (float*)loss);
  if (fwd_split(B) == 2) {
    hipLaunchKernelGGL(net_fused_fwd_split_kernel, dim3(2 * B), dim3(256),
                       0, S(stream), (const float*)x, (const float*)w1,
                       (const float*)b1, (const float*)w2,
                       (const float*)b2, (const float*)wf1,
                       (const float*)bf1, (const float*)wf2,
                       (const float*)bf2, (const int64_t*)tgt,
                       (float*)p1_ws, (uint8_t*)idx1_ws, (uint8_t*)m2_ws,
                       (float*)p2_ws, (uint8_t*)idx2_ws, (float*)h1_ws,
                       (uint8_t*)m3_ws, (float*)d3_ws, (float*)logp_ws,
                       (float*)loss, (float*)loss_part,
                       (const unsigned long long*)seed_dev, B,
                       training ? 1 : 0, fwd_flags_buf(S(stream)),
                       fwd_token());
    return;
  }
  hipLaunchKernelGGL(net_fused_fwd_kernel, dim3(grid_for(B, 1)), dim3(256),
                     0, S(stream), (const float*)x, (const float*)w1,
                     (const float*)b1, (const float*)w2, (const float*)b2,
                     (const float*)wf1, (const float*)bf1,
                     (const float*)wf2, (const float*)bf2,
                     (const int64_t*)tgt, (float*)p1_ws, (uint8_t*)idx1_ws,
                     (uint8_t*)m2_ws, (float*)p2_ws, (uint8_t*)idx2_ws,
                     (float*)h1_ws, (uint8_t*)m3_ws, (float*)d3_ws,
                     (float*)logp_ws, (float*)loss, (float*)loss_part,
                     (const unsigned long long*)seed_dev, B,
                     training ? 1 : 0);
}

static void launch_linear_gw(uintptr_t xp, uintptr_t gyp, uintptr_t gw,
                             uintptr_t gb, int B, int K, int N,
                             uintptr_t stream) {
  const int wn = N * K;
  const int wtiles = (wn + BLK - 1) / BLK;
  int bchunk = B;
  while (bchunk > 1 && wtiles * ((B + bchunk - 1) / bchunk) < 512)
    bchunk = (bchunk + 1) / 2;
  const int nchunks = (B + bchunk - 1) / bchunk;
  if (nchunks > 1) {
    HIP_CHECK(hipMemsetAsync((void*)gw, 0, (size_t)wn * sizeof(float),
                             S(stream)));
    if (gb)
      HIP_CHECK(hipMemsetAsync((void*)gb, 0, N * sizeof(float), S(stream)));
  }
  hipLaunchKernelGGL(linear_bwd_w_kernel, dim3(wtiles, nchunks), dim3(BLK),
                     0, S(stream), (const float*)xp, (const float*)gyp,
                     (const float*)nullptr, (float*)gw, (float*)gb, B, K,
                     N, bchunk);
}

static void launch_conv_gw(uintptr_t xp, uintptr_t gyp, uintptr_t gw,
                           uintptr_t gb, int B, int C, int H, int W, int K,
                           uintptr_t stream) {
  const int wn = K * C * 25;
  const int wtiles = (wn + BLK - 1) / BLK;
  int bchunk = B;
  while (bchunk > 1 && wtiles * ((B + bchunk - 1) / bchunk) < 512)
    bchunk = (bchunk + 1) / 2;
  const int nchunks = (B + bchunk - 1) / bchunk;
  if (nchunks > 1) {
    HIP_CHECK(hipMemsetAsync((void*)gw, 0, (size_t)wn * sizeof(float),
                             S(stream)));
    if (gb)
      HIP_CHECK(hipMemsetAsync((void*)gb, 0, K * sizeof(float), S(stream)));
  }
  hipLaunchKernelGGL(conv2d_bwd_w_kernel, dim3(wtiles, nchunks), dim3(BLK),
                     0, S(stream), (const float*)xp, (const float*)gyp,
                     (float*)gw, (float*)gb, B, C, H, W, K, 5, 5, bchunk);
}

void net_fused_bwd(uintptr_t x, uintptr_t w2, uintptr_t wf1, uintptr_t wf2,
                   uintptr_t tgt, uintptr_t gl,
                   uintptr_t p1_ws, uintptr_t idx1_ws, uintptr_t m2_ws,
                   uintptr_t p2_ws, uintptr_t idx2_ws, uintptr_t h1_ws,
                   uintptr_t m3_ws, uintptr_t d3_ws, uintptr_t logp_ws,
                   uintptr_t glog_ws, uintptr_t gh1_ws, uintptr_t ga2_ws,
                   uintptr_t ga1_ws, uintptr_t part_ws,
                   uintptr_t gw1, uintptr_t gb1, uintptr_t gw2,
                   uintptr_t gb2, uintptr_t gwf1, uintptr_t gbf1,
                   uintptr_t gwf2, uintptr_t gbf2, int B, bool training,
                   uintptr_t loss_part, uintptr_t loss_out,
                   uintptr_t seed_dev, uintptr_t stream) {
  // enough sibling workgroups per sample to fill the 256 CUs
  // (DTP_BWD_SPLIT=1/2/4/8 overrides, for microbenchmarks)
  int split = 0;
  if (const char* e = std::getenv("DTP_BWD_SPLIT")) split = std::atoi(e);
  if (split != 1 && split != 2 && split != 4 && split != 8) {
    // microbench (profiles/): with the 4-wide-blocked hot loop,
    // split=2 at B=128 (256 workgroups, 180 strips each) is fastest
    split = 1;
    while (split < 8 && B * split < 256) split *= 2;
  }
  const int nblk = grid_for((int64_t)B * split, 1);
  hipLaunchKernelGGL(net_fused_bwd_kernel, dim3(nblk), dim3(256),
                     0, S(stream), (const float*)w2, (const float*)wf1,
                     (const float*)wf2, (const int64_t*)tgt,
                     (const float*)gl, (const uint8_t*)idx1_ws,
                     (const uint8_t*)m2_ws, (const uint8_t*)idx2_ws,
                     (const float*)h1_ws, (const uint8_t*)m3_ws,
                     (const float*)logp_ws, (float*)glog_ws,
                     (float*)gh1_ws, (float*)ga2_ws, (float*)ga1_ws, B,
                     training ? 1 : 0, split);
  bool adaptive = false;
  if (const char* e = std::getenv("DTP_GW_ADAPTIVE"))
    adaptive = std::atoi(e) != 0;
  if (adaptive) {
    // legacy path kept for comparison: per-op adaptive chunked
    // reductions.  Measured 2.85 ms/step at B=4096 vs ~0.7 ms for the
    // partial scheme (profiles/) — not the default at any batch size.
    if (loss_part)
      throw std::runtime_error("net_fused_bwd: loss_part mode needs "
                               "the partial gw path");
    launch_conv_gw(x, ga1_ws, gw1, gb1, B, 1, 28, 28, N_C1K, stream);
    launch_conv_gw(p1_ws, ga2_ws, gw2, gb2, B, 10, 12, 12, N_C2K,
                   stream);
    launch_linear_gw(p2_ws, gh1_ws, gwf1, gbf1, B, N_P2, N_H1, stream);
    launch_linear_gw(d3_ws, glog_ws, gwf2, gbf2, B, N_H1, N_CLS, stream);
    return;
  }
  // one segmented partial kernel over <=32 batch chunks + one combine
  // (no memsets, no atomics); any batch size.  DTP_GW_NCH overrides
  // the chunk count (fewer chunks = less partial traffic for the
  // combine, more batch per partial block — sweep on hardware).
  const int nch = gw_nch(B);
  const int bchunk = (B + nch - 1) / nch;
  // fold path keeps uniform chunks (its per-column tickets assume it)
  const int nch1 = gw_fold_on() ? nch : gw_nch1(B, nch);
  const int nch2 = gw_fold_on() ? nch : gw_nch2(B, nch);
  const int bchunk1 = (B + nch1 - 1) / nch1;
  const int bchunk2 = (B + nch2 - 1) / nch2;
  const int gw_gy = std::max(nch, std::max(nch1, nch2));
  GwPtrs gp;
  gp.p[0] = (float*)gw1; gp.p[1] = (float*)gb1;
  gp.p[2] = (float*)gw2; gp.p[3] = (float*)gb2;
  gp.p[4] = (float*)gwf1; gp.p[5] = (float*)gbf1;
  gp.p[6] = (float*)gwf2; gp.p[7] = (float*)gbf2;
  unsigned long long* sb = (training && seed_dev)
      ? (unsigned long long*)seed_dev : nullptr;
  if (gw_fold_on()) {
    GwPtrs none{};
    hipLaunchKernelGGL(net_gw_partial_fold_kernel<false>,
                       dim3(GW_TILES, nch), dim3(256), 0, S(stream),
                       gw_c1_subs(bchunk), (const float*)x,
                       (const float*)p1_ws, (const float*)p2_ws,
                       (const float*)d3_ws, (const float*)ga1_ws,
                       (const float*)ga2_ws, (const float*)gh1_ws,
                       (const float*)glog_ws, (float*)part_ws, B, bchunk,
                       gp, none, none, 0.f, 0.f,
                       (const float*)loss_part, (float*)loss_out,
                       fwd_grid(B), sb, gw_cnt_buf(S(stream)));
    return;
  }
  hipLaunchKernelGGL(net_gw_partial_kernel, dim3(GW_TILES, gw_gy),
                     dim3(256), 0, S(stream), gw_c1_subs(bchunk),
                     (const float*)x,
                     (const float*)p1_ws, (const float*)p2_ws,
                     (const float*)d3_ws, (const float*)ga1_ws,
                     (const float*)ga2_ws, (const float*)gh1_ws,
                     (const float*)glog_ws, (float*)part_ws, B, bchunk,
                     bchunk1, bchunk2, nch, nch1, nch2, 0);
  hipLaunchKernelGGL(net_gw_combine_kernel,
                     dim3((GW_TOTAL * 4 + 255) / 256), dim3(256), 0,
                     S(stream), (const float*)part_ws, gp, nch, nch1,
                     nch2, gw_c1_subs(bchunk) - 1,
                     (const float*)loss_part, (float*)loss_out,
                     fwd_grid(B), sb);
}

// Combined fwd+bwd (one dispatch) + gw partial + combine: a 3-dispatch
// training step (vs 4 for fwd / bwd / partial / combine).  When prm_v
// is non-empty the combine also applies the SGD+momentum update
// (single-GPU: 3 dispatches total including the optimizer).
void net_fused_fwdbwd(
    uintptr_t x, uintptr_t w1, uintptr_t b1, uintptr_t w2, uintptr_t b2,
    uintptr_t wf1, uintptr_t bf1, uintptr_t wf2, uintptr_t bf2,
    uintptr_t tgt,
    uintptr_t p1_ws, uintptr_t idx1_ws, uintptr_t m2_ws, uintptr_t p2_ws,
    uintptr_t idx2_ws, uintptr_t h1_ws, uintptr_t m3_ws, uintptr_t d3_ws,
    uintptr_t logp_ws, uintptr_t glog_ws, uintptr_t gh1_ws,
    uintptr_t ga2_ws, uintptr_t ga1_ws, uintptr_t part_ws,
    const std::vector<uintptr_t>& grd_v,
    const std::vector<uintptr_t>& prm_v,
    const std::vector<uintptr_t>& buf_v,
    double lr, double mu, int B, bool training,
    uintptr_t loss_part, uintptr_t loss_out, uintptr_t seed_dev,
    uintptr_t stream) {
  if (grd_v.size() != 8)
    throw std::runtime_error("net_fused_fwdbwd: expected 8 grad ptrs");
  const int nblk = grid_for(B, 1);
  hipLaunchKernelGGL(net_fused_fwdbwd_kernel, dim3(nblk), dim3(256), 0,
                     S(stream), (const float*)x, (const float*)w1,
                     (const float*)b1, (const float*)w2, (const float*)b2,
                     (const float*)wf1, (const float*)bf1,
                     (const float*)wf2, (const float*)bf2,
                     (const int64_t*)tgt, (float*)p1_ws,
                     (uint8_t*)idx1_ws, (uint8_t*)m2_ws, (float*)p2_ws,
                     (uint8_t*)idx2_ws, (float*)h1_ws, (uint8_t*)m3_ws,
                     (float*)d3_ws, (float*)logp_ws, (float*)glog_ws,
                     (float*)gh1_ws, (float*)ga2_ws, (float*)ga1_ws,
                     (float*)loss_part,
                     (const unsigned long long*)seed_dev, B,
                     training ? 1 : 0);
  const int nch = gw_nch(B);
  const int bchunk = (B + nch - 1) / nch;
  // fold path keeps uniform chunks (its per-column tickets assume it)
  const int nch1 = gw_fold_on() ? nch : gw_nch1(B, nch);
  const int nch2 = gw_fold_on() ? nch : gw_nch2(B, nch);
  const int bchunk1 = (B + nch1 - 1) / nch1;
  const int bchunk2 = (B + nch2 - 1) / nch2;
  const int gw_gy = std::max(nch, std::max(nch1, nch2));
  GwPtrs gp{}, pp{}, bp{};
  for (int i = 0; i < 8; ++i) gp.p[i] = (float*)grd_v[i];
  const bool sgd = !prm_v.empty();
  if (sgd) {
    for (int i = 0; i < 8; ++i) {
      pp.p[i] = (float*)prm_v[i];
      bp.p[i] = (i < (int)buf_v.size()) ? (float*)buf_v[i] : nullptr;
    }
  }
  unsigned long long* sb = (training && seed_dev)
      ? (unsigned long long*)seed_dev : nullptr;
  if (gw_fold_on()) {
    auto* kern = sgd ? net_gw_partial_fold_kernel<true>
                     : net_gw_partial_fold_kernel<false>;
    hipLaunchKernelGGL(kern, dim3(GW_TILES, nch), dim3(256), 0, S(stream),
                       gw_c1_subs(bchunk), (const float*)x,
                       (const float*)p1_ws, (const float*)p2_ws,
                       (const float*)d3_ws, (const float*)ga1_ws,
                       (const float*)ga2_ws, (const float*)gh1_ws,
                       (const float*)glog_ws, (float*)part_ws, B, bchunk,
                       gp, pp, bp, (float)lr, (float)mu,
                       (const float*)loss_part, (float*)loss_out, nblk,
                       sb, gw_cnt_buf(S(stream)));
    return;
  }
  hipLaunchKernelGGL(net_gw_partial_kernel, dim3(GW_TILES, gw_gy),
                     dim3(256), 0, S(stream), gw_c1_subs(bchunk),
                     (const float*)x,
                     (const float*)p1_ws, (const float*)p2_ws,
                     (const float*)d3_ws, (const float*)ga1_ws,
                     (const float*)ga2_ws, (const float*)gh1_ws,
                     (const float*)glog_ws, (float*)part_ws, B, bchunk,
                     bchunk1, bchunk2, nch, nch1, nch2, 0);
  if (!sgd) {
    hipLaunchKernelGGL(net_gw_combine_kernel,
                       dim3((GW_TOTAL * 4 + 255) / 256), dim3(256), 0,
                       S(stream), (const float*)part_ws, gp, nch,
                       nch1, nch2, gw_c1_subs(bchunk) - 1,
                       (const float*)loss_part, (float*)loss_out,
                       nblk, sb);
  } else {
    hipLaunchKernelGGL(net_gw_combine_sgd_kernel,
                       dim3((GW_TOTAL * 4 + 255) / 256), dim3(256), 0,
                       S(stream), (const float*)part_ws, gp, pp, bp,
                       nch, nch1, nch2, gw_c1_subs(bchunk) - 1,
                       (float)lr, (float)mu, (const float*)loss_part,
                       (float*)loss_out, nblk, sb);
  }
}

// net_fused_bwd + the optimizer update fused into the combine kernel
// (single-GPU path: no all-reduce between combine and step).  The
// segmented-partial weight-gradient path handles any batch size.
void net_fused_bwd_sgd(uintptr_t x, uintptr_t w2, uintptr_t wf1,
                       uintptr_t wf2, uintptr_t tgt, uintptr_t gl,
                       uintptr_t p1_ws, uintptr_t idx1_ws,
                       uintptr_t m2_ws, uintptr_t p2_ws,
                       uintptr_t idx2_ws, uintptr_t h1_ws,
                       uintptr_t m3_ws, uintptr_t d3_ws,
                       uintptr_t logp_ws, uintptr_t glog_ws,
                       uintptr_t gh1_ws, uintptr_t ga2_ws,
                       uintptr_t ga1_ws, uintptr_t part_ws,
                       const std::vector<uintptr_t>& grd_v,
                       const std::vector<uintptr_t>& prm_v,
                       const std::vector<uintptr_t>& buf_v,
                       double lr, double mu, int B, bool training,
                       uintptr_t loss_part, uintptr_t loss_out,
                       uintptr_t seed_dev, uintptr_t stream) {
  if (grd_v.size() != 8 || prm_v.size() != 8)
    throw std::runtime_error("net_fused_bwd_sgd: expected 8 pointers");
  int split = 0;
  if (const char* e = std::getenv("DTP_BWD_SPLIT")) split = std::atoi(e);
  if (split != 1 && split != 2 && split != 4 && split != 8) {
    split = 1;
    while (split < 8 && B * split < 256) split *= 2;
  }
  const int nblk = grid_for((int64_t)B * split, 1);
  hipLaunchKernelGGL(net_fused_bwd_kernel, dim3(nblk), dim3(256),
                     0, S(stream), (const float*)w2, (const float*)wf1,
                     (const float*)wf2, (const int64_t*)tgt,
                     (const float*)gl, (const uint8_t*)idx1_ws,
                     (const uint8_t*)m2_ws, (const uint8_t*)idx2_ws,
                     (const float*)h1_ws, (const uint8_t*)m3_ws,
                     (const float*)logp_ws, (float*)glog_ws,
                     (float*)gh1_ws, (float*)ga2_ws, (float*)ga1_ws, B,
                     training ? 1 : 0, split);
  const int nch = gw_nch(B);
  const int bchunk = (B + nch - 1) / nch;
  // fold path keeps uniform chunks (its per-column tickets assume it)
  const int nch1 = gw_fold_on() ? nch : gw_nch1(B, nch);
  const int nch2 = gw_fold_on() ? nch : gw_nch2(B, nch);
  const int bchunk1 = (B + nch1 - 1) / nch1;
  const int bchunk2 = (B + nch2 - 1) / nch2;
  const int gw_gy = std::max(nch, std::max(nch1, nch2));
  GwPtrs gp{}, pp{}, bp{};
  for (int i = 0; i < 8; ++i) {
    gp.p[i] = (float*)grd_v[i];
    pp.p[i] = (float*)prm_v[i];
    bp.p[i] = (i < (int)buf_v.size()) ? (float*)buf_v[i] : nullptr;
  }
  unsigned long long* sb = (training && seed_dev)
      ? (unsigned long long*)seed_dev : nullptr;
  if (gw_fold_on()) {
    hipLaunchKernelGGL(net_gw_partial_fold_kernel<true>,
                       dim3(GW_TILES, nch), dim3(256), 0, S(stream),
                       gw_c1_subs(bchunk), (const float*)x,
                       (const float*)p1_ws, (const float*)p2_ws,
                       (const float*)d3_ws, (const float*)ga1_ws,
                       (const float*)ga2_ws, (const float*)gh1_ws,
                       (const float*)glog_ws, (float*)part_ws, B, bchunk,
                       gp, pp, bp, (float)lr, (float)mu,
                       (const float*)loss_part, (float*)loss_out,
                       fwd_grid(B), sb, gw_cnt_buf(S(stream)));
    return;
  }
  hipLaunchKernelGGL(net_gw_partial_kernel, dim3(GW_TILES, gw_gy),
                     dim3(256), 0, S(stream), gw_c1_subs(bchunk),
                     (const float*)x,
                     (const float*)p1_ws, (const float*)p2_ws,
                     (const float*)d3_ws, (const float*)ga1_ws,
                     (const float*)ga2_ws, (const float*)gh1_ws,
                     (const float*)glog_ws, (float*)part_ws, B, bchunk,
                     bchunk1, bchunk2, nch, nch1, nch2, 0);
  hipLaunchKernelGGL(net_gw_combine_sgd_kernel,
                     dim3((GW_TOTAL * 4 + 255) / 256), dim3(256), 0,
                     S(stream), (const float*)part_ws, gp, pp, bp,
                     nch, nch1, nch2, gw_c1_subs(bchunk) - 1,
                     (float)lr, (float)mu, (const float*)loss_part,
                     (float*)loss_out, fwd_grid(B), sb);
}

// raw combine launch (microbenchmarks: time the combine/sgd dispatch
// in isolation)
void net_gw_combine_raw(uintptr_t part_ws,
                        const std::vector<uintptr_t>& grd_v, int nch,
                        int nch1, int nch2, int c1_ext,
                        uintptr_t stream) {
  GwPtrs gp{};
  for (int i = 0; i < 8; ++i) gp.p[i] = (float*)grd_v[i];
  hipLaunchKernelGGL(net_gw_combine_kernel,
                     dim3((GW_TOTAL * 4 + 255) / 256), dim3(256), 0,
                     S(stream), (const float*)part_ws, gp, nch, nch1,
                     nch2, c1_ext, nullptr, nullptr, 0, nullptr);
}

// raw combine+sgd launch (microbenchmarks)
void net_gw_combine_sgd_raw(uintptr_t part_ws,
                            const std::vector<uintptr_t>& grd_v,
                            const std::vector<uintptr_t>& prm_v,
                            const std::vector<uintptr_t>& buf_v,
                            int nch, int nch1, int nch2, int c1_ext,
                            double lr,
                            double mu,
                            uintptr_t loss_part, uintptr_t loss_out,
                            int nblk_fwd, uintptr_t stream) {
  GwPtrs gp{}, pp{}, bp{};
  for (int i = 0; i < 8; ++i) {
    gp.p[i] = (float*)grd_v[i];
    pp.p[i] = (float*)prm_v[i];
    bp.p[i] = (i < (int)buf_v.size()) ? (float*)buf_v[i] : nullptr;
  }
  hipLaunchKernelGGL(net_gw_combine_sgd_kernel,
                     dim3((GW_TOTAL * 4 + 255) / 256), dim3(256), 0,
                     S(stream), (const float*)part_ws, gp, pp, bp, nch,
                     nch1, nch2, c1_ext, (float)lr, (float)mu,
                     (const float*)loss_part, (float*)loss_out,
                     nblk_fwd, nullptr);
}

// raw tile-segment launch of the partial weight-gradient kernel
// (microbenchmarks: time [conv2 | fc1 | conv1 | fc2] separately)
void net_gw_partial_raw(uintptr_t x, uintptr_t p1_ws, uintptr_t p2_ws,
                        uintptr_t d3_ws, uintptr_t ga1_ws,
                        uintptr_t ga2_ws, uintptr_t gh1_ws,
                        uintptr_t glog_ws, uintptr_t part_ws, int B,
                        int bchunk, int bchunk1, int bchunk2,
                        int tile_base, int ntiles, int nch, int nch1,
                        int nch2, uintptr_t stream) {
  const int gy = std::max(nch, std::max(nch1, nch2));
  hipLaunchKernelGGL(net_gw_partial_kernel, dim3(ntiles, gy), dim3(256),
                     0, S(stream), gw_c1_subs(bchunk), (const float*)x,
                     (const float*)p1_ws, (const float*)p2_ws,
                     (const float*)d3_ws, (const float*)ga1_ws,
                     (const float*)ga2_ws, (const float*)gh1_ws,
                     (const float*)glog_ws, (float*)part_ws, B, bchunk,
                     bchunk1, bchunk2, nch, nch1, nch2, tile_base);
}

// cooperative grid-barrier cost probe: `nsync` grid.sync()s and
// nothing else, at fwd-kernel-like occupancy (32 KB LDS, 256 thr).
// Times the lever that decides whether mid-kernel sibling exchanges
// (fwd split, SURVEY-future) can beat kernel boundaries.
__global__ void __launch_bounds__(256)
barrier_probe_kernel(int nsync, float* sink) {
  __shared__ float smem[8192];
  smem[threadIdx.x] = (float)threadIdx.x;
  cg::grid_group grid = cg::this_grid();
  for (int i = 0; i < nsync; ++i) grid.sync();
  if (threadIdx.x == 0 && blockIdx.x == 0) sink[0] = smem[0];
}

void barrier_probe(int nblk, int nsync, uintptr_t sink, uintptr_t stream) {
  int dev = 0;
  HIP_CHECK(hipGetDevice(&dev));
  int nsync_v = nsync;
  float* sinkp = (float*)sink;
  void* args[] = {&nsync_v, &sinkp};
  HIP_CHECK(hipLaunchCooperativeKernel(
      reinterpret_cast<const void*>(barrier_probe_kernel), dim3(nblk),
      dim3(256), args, 0, S(stream)));
}

// hand-rolled resident-grid flag barrier probe (the guide's G16
// last-arriver pattern: agent-scope acq/rel on a generation counter,
// relaxed poll).  Launched cooperatively ONLY for the co-residency
// guarantee; measures what a grid barrier costs without
// cooperative-groups' sync machinery (17-42 us, see barrier_probe).
__global__ void __launch_bounds__(256)
flag_barrier_probe_kernel(int nsync, unsigned int* bar, float* sink) {
  __shared__ float smem[8192];
  smem[threadIdx.x] = (float)threadIdx.x;
  for (int i = 0; i < nsync; ++i) {
    __syncthreads();
    if (threadIdx.x == 0) {
      const unsigned int gen = __hip_atomic_load(
          &bar[1], __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_AGENT);
      const unsigned int arrived =
          __hip_atomic_fetch_add(&bar[0], 1u, __ATOMIC_ACQ_REL,
                                 __HIP_MEMORY_SCOPE_AGENT) + 1;
      if (arrived == gridDim.x) {
        __hip_atomic_store(&bar[0], 0u, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
        __hip_atomic_fetch_add(&bar[1], 1u, __ATOMIC_RELEASE,
                               __HIP_MEMORY_SCOPE_AGENT);
      } else {
        while (__hip_atomic_load(&bar[1], __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_AGENT) == gen)
          __builtin_amdgcn_s_sleep(8);
        (void)__hip_atomic_load(&bar[1], __ATOMIC_ACQUIRE,
                                __HIP_MEMORY_SCOPE_AGENT);
      }
    }
    __syncthreads();
  }
  if (threadIdx.x == 0 && blockIdx.x == 0) sink[0] = smem[0];
}

void flag_barrier_probe(int nblk, int nsync, uintptr_t bar,
                        uintptr_t sink, uintptr_t stream) {
  int nsync_v = nsync;
  unsigned int* barp = (unsigned int*)bar;
  float* sinkp = (float*)sink;
  void* args[] = {&nsync_v, &barp, &sinkp};
  HIP_CHECK(hipLaunchCooperativeKernel(
      reinterpret_cast<const void*>(flag_barrier_probe_kernel),
      dim3(nblk), dim3(256), args, 0, S(stream)));
}

// ---- single-launch training step (cooperative) --------------------------
int net_step_max_blocks() {
  static int cached = -2;
  if (cached != -2) return cached;
  int dev = 0;
  if (hipGetDevice(&dev) != hipSuccess) { cached = 0; return 0; }
  int coop = 0;
  if (hipDeviceGetAttribute(&coop, hipDeviceAttributeCooperativeLaunch,
                            dev) != hipSuccess || !coop) {
    cached = 0;
    return 0;
  }
  int per_cu = 0;
  if (hipOccupancyMaxActiveBlocksPerMultiprocessor(
          &per_cu, reinterpret_cast<const void*>(net_step_kernel), 256,
          0) != hipSuccess || per_cu < 1) {
    cached = 0;
    return 0;
  }
  hipDeviceProp_t prop;
  if (hipGetDeviceProperties(&prop, dev) != hipSuccess) {
    cached = 0;
    return 0;
  }
  int nb = per_cu * prop.multiProcessorCount;
  if (nb > 512) nb = 512;  // loss_part reduction handles <= 512
  cached = nb;
  return nb;
}

bool net_step_available() { return net_step_max_blocks() > 0; }

void net_step(uintptr_t x, uintptr_t tgt, uintptr_t gl,
              uintptr_t p1_ws, uintptr_t idx1_ws, uintptr_t m2_ws,
              uintptr_t p2_ws, uintptr_t idx2_ws, uintptr_t h1_ws,
              uintptr_t m3_ws, uintptr_t d3_ws, uintptr_t logp_ws,
              uintptr_t glog_ws, uintptr_t gh1_ws, uintptr_t ga2_ws,
              uintptr_t ga1_ws, uintptr_t part_ws, uintptr_t loss_part,
              uintptr_t loss_out, uintptr_t seed_dev,
              const std::vector<uintptr_t>& prm_v,
              const std::vector<uintptr_t>& grd_v,
              const std::vector<uintptr_t>& buf_v,
              double lr, double mu, bool do_sgd, int B, bool training,
              uintptr_t stream) {
  const int nblk = net_step_max_blocks();
  if (nblk < 1)
    throw std::runtime_error(
        "net_step: cooperative launch unavailable on this device");
  if (prm_v.size() != 8 || grd_v.size() != 8)
    throw std::runtime_error("net_step: expected 8 param/grad pointers");
  int split = 1;
  while (split < 8 && B * split < nblk) split *= 2;
  int bchunk = (B + 31) / 32;
  int nch = (B + bchunk - 1) / bchunk;
  GwPtrs prm{}, grd{}, buf{};
  for (int i = 0; i < 8; ++i) {
    prm.p[i] = (float*)prm_v[i];
    grd.p[i] = (float*)grd_v[i];
    buf.p[i] = (i < (int)buf_v.size()) ? (float*)buf_v[i] : nullptr;
  }
  const float* xp = (const float*)x;
  const int64_t* tgtp = (const int64_t*)tgt;
  const float* glp = (const float*)gl;
  float* p1p = (float*)p1_ws;
  uint8_t* idx1p = (uint8_t*)idx1_ws;
  uint8_t* m2p = (uint8_t*)m2_ws;
  float* p2p = (float*)p2_ws;
  uint8_t* idx2p = (uint8_t*)idx2_ws;
  float* h1p = (float*)h1_ws;
  uint8_t* m3p = (uint8_t*)m3_ws;
  float* d3p = (float*)d3_ws;
  float* logpp = (float*)logp_ws;
  float* glogp = (float*)glog_ws;
  float* gh1p = (float*)gh1_ws;
  float* ga2p = (float*)ga2_ws;
  float* ga1p = (float*)ga1_ws;
  float* partp = (float*)part_ws;
  float* lpartp = (float*)loss_part;
  float* loutp = (float*)loss_out;
  unsigned long long* seedp = (unsigned long long*)seed_dev;
  float lrf = (float)lr, muf = (float)mu;
  int do_sgd_i = do_sgd ? 1 : 0, Bi = B, tri = training ? 1 : 0;
  void* args[] = {&xp, &tgtp, &glp, &p1p, &idx1p, &m2p, &p2p, &idx2p,
                  &h1p, &m3p, &d3p, &logpp, &glogp, &gh1p, &ga2p, &ga1p,
                  &partp, &lpartp, &loutp, &seedp, &prm, &grd, &buf,
                  &lrf, &muf, &do_sgd_i, &Bi, &tri, &split, &bchunk,
                  &nch};
  HIP_CHECK(hipLaunchCooperativeKernel(
      reinterpret_cast<const void*>(net_step_kernel), dim3(nblk),
      dim3(256), args, 0, S(stream)));
}

void add_inplace(uintptr_t dst, uintptr_t src, int64_t n, int dtype,
                 uintptr_t stream) {
  if (dtype == 7) {  // ncclFloat32 numbering (dist wrapper's _DTYPE)
    hipLaunchKernelGGL(add_inplace_f32_kernel, dim3(grid_for(n, BLK, 4)),
                       dim3(BLK), 0, S(stream), (float*)dst,
                       (const float*)src, n);
  } else if (dtype == 9) {  // bf16
    hipLaunchKernelGGL(add_inplace_bf16_kernel, dim3(grid_for(n, BLK, 8)),
                       dim3(BLK), 0, S(stream), (__hip_bfloat16*)dst,
                       (const __hip_bfloat16*)src, n);
  } else {
    throw std::runtime_error("add_inplace: unsupported dtype");
  }
}

void reduce_columns(uintptr_t dst, uintptr_t src, int P, int64_t stride,
                    int64_t n, double scale, int dtype, uintptr_t stream) {
  if (dtype == 7) {
    hipLaunchKernelGGL(reduce_columns_f32_kernel,
                       dim3(grid_for(n, BLK, 4)), dim3(BLK), 0, S(stream),
                       (float*)dst, (const float*)src, P, stride, n,
                       (float)scale);
  } else if (dtype == 9) {
    hipLaunchKernelGGL(reduce_columns_bf16_kernel,
                       dim3(grid_for(n, BLK, 8)), dim3(BLK), 0, S(stream),
                       (__hip_bfloat16*)dst, (const __hip_bfloat16*)src, P,
                       stride, n, (float)scale);
  } else {
    throw std::runtime_error("reduce_columns: unsupported dtype");
  }
}

void scale_f32(uintptr_t dst, double s, int64_t n, uintptr_t stream) {
  hipLaunchKernelGGL(scale_f32_kernel, dim3(grid_for(n, BLK, 4)), dim3(BLK),
                     0, S(stream), (float*)dst, (float)s, n);
}

void copy_throttled(uintptr_t dst, uintptr_t src, int64_t n, int nblocks,
                    uintptr_t stream) {
  hipLaunchKernelGGL(copy_throttled_f32_kernel, dim3(nblocks), dim3(BLK),
                     0, S(stream), (float*)dst, (const float*)src, n);
}

}  // namespace

PYBIND11_MODULE(_kernels, m) {
  m.doc() = "CDNA4 HIP kernels for the dist_tuto_pth_amd training path";
  m.def("conv2d_fwd", &conv2d_fwd);
  m.def("conv2d_bwd", &conv2d_bwd);
  m.def("maxpool2d_relu_fwd", &maxpool2d_relu_fwd);
  m.def("maxpool2d_relu_bwd", &maxpool2d_relu_bwd);
  m.def("relu_fwd", &relu_fwd);
  m.def("relu_bwd", &relu_bwd);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("dropout2d_fwd", &dropout2d_fwd);
  m.def("dropout2d_bwd", &dropout2d_bwd);
  m.def("linear_fwd", &linear_fwd);
  m.def("linear_bwd", &linear_bwd);
  m.def("log_softmax_fwd", &log_softmax_fwd);
  m.def("log_softmax_bwd", &log_softmax_bwd);
  m.def("nll_loss_fwd", &nll_loss_fwd);
  m.def("nll_loss_bwd", &nll_loss_bwd);
  m.def("log_softmax_nll_fwd", &log_softmax_nll_fwd);
  m.def("log_softmax_nll_bwd", &log_softmax_nll_bwd);
  m.def("sgd_step", &sgd_step);
  m.def("net_fused_fwd", &net_fused_fwd);
  m.def("net_fused_bwd", &net_fused_bwd);
  m.def("net_fused_fwdbwd", &net_fused_fwdbwd);
  m.def("net_step", &net_step);
  m.def("net_step_available", &net_step_available);
  m.def("net_gw_partial_raw", &net_gw_partial_raw);
  m.def("net_gw_combine_raw", &net_gw_combine_raw);
  m.def("net_gw_combine_sgd_raw", &net_gw_combine_sgd_raw);
  m.def("barrier_probe", &barrier_probe);
  m.def("flag_barrier_probe", &flag_barrier_probe);
  m.def("net_fused_bwd_sgd", &net_fused_bwd_sgd);
  m.def("add_inplace", &add_inplace);
  m.def("copy_throttled", &copy_throttled);
  m.def("reduce_columns", &reduce_columns);
  m.def("scale_f32", &scale_f32);
}
