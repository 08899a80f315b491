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

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
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
                                  int K, int R, int S_) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  const int OH = H - R + 1, OW = W - S_ + 1;
  const int xn = C * H * W;
  const int wn = K * C * R * S_;
  float* xs = smem;        // [C*H*W]
  float* ws = smem + xn;   // [K*C*R*S]

  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    // stage input plane + weights
    for (int i = threadIdx.x; i < xn; i += blockDim.x)
      xs[i] = x[(int64_t)b * xn + i];
    for (int i = threadIdx.x; i < wn; i += blockDim.x)
      ws[i] = w[i];
    __syncthreads();

    const int on = K * OH * OW;
    for (int i = threadIdx.x; i < on; i += blockDim.x) {
      const int k = i / (OH * OW);
      const int oh = (i / OW) % OH;
      const int ow = i % OW;
      float acc = bias ? bias[k] : 0.f;
      const float* wk = ws + k * C * R * S_;
      for (int c = 0; c < C; ++c) {
        const float* xc = xs + c * H * W;
        const float* wc = wk + c * R * S_;
        #pragma unroll 5
        for (int r = 0; r < R; ++r) {
          const float* xrow = xc + (oh + r) * W + ow;
          const float* wrow = wc + r * S_;
          float a = 0.f;
          #pragma unroll 5
          for (int s = 0; s < S_; ++s) a += xrow[s] * wrow[s];
          acc += a;
        }
      }
      out[(int64_t)b * on + i] = acc;
    }
    __syncthreads();
  }
}

// backward dx: gx[b,c,h,w] = sum_k sum_{r,s} gy[b,k,h-r,w-s] * w[k,c,r,s]
// (valid range only).  gy plane (K*OH*OW <= 5.76 KB) + weights in LDS.
__global__ void conv2d_bwd_x_kernel(const float* __restrict__ gy,
                                    const float* __restrict__ w,
                                    float* __restrict__ gx,
                                    int B, int C, int H, int W,
                                    int K, int R, int S_) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  const int OH = H - R + 1, OW = W - S_ + 1;
  const int gn = K * OH * OW;
  const int wn = K * C * R * S_;
  float* gys = smem;
  float* ws = smem + gn;

  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    for (int i = threadIdx.x; i < gn; i += blockDim.x)
      gys[i] = gy[(int64_t)b * gn + i];
    for (int i = threadIdx.x; i < wn; i += blockDim.x)
      ws[i] = w[i];
    __syncthreads();

    const int xn = C * H * W;
    for (int i = threadIdx.x; i < xn; i += blockDim.x) {
      const int c = i / (H * W);
      const int h = (i / W) % H;
      const int wcol = i % W;
      float acc = 0.f;
      for (int k = 0; k < K; ++k) {
        const float* gk = gys + k * OH * OW;
        const float* wk = ws + (k * C + c) * R * S_;
        #pragma unroll 5
        for (int r = 0; r < R; ++r) {
          const int oh = h - r;
          if (oh < 0 || oh >= OH) continue;
          #pragma unroll 5
          for (int s = 0; s < S_; ++s) {
            const int ow = wcol - s;
            if (ow < 0 || ow >= OW) continue;
            acc += gk[oh * OW + ow] * wk[r * S_ + s];
          }
        }
      }
      gx[(int64_t)b * xn + i] = acc;
    }
    __syncthreads();
  }
}

// backward dw/db: per-block (per batch element) partials accumulated in
// LDS, then atomically added into gw/gb (gw zeroed by the caller).
__global__ void conv2d_bwd_w_kernel(const float* __restrict__ x,
                                    const float* __restrict__ gy,
                                    float* __restrict__ gw,
                                    float* __restrict__ gb,
                                    int B, int C, int H, int W,
                                    int K, int R, int S_) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  const int OH = H - R + 1, OW = W - S_ + 1;
  const int xn = C * H * W;
  const int gn = K * OH * OW;
  const int wn = K * C * R * S_;
  float* xs = smem;            // [xn]
  float* gys = smem + xn;      // [gn]
  float* wacc = gys + gn;      // [wn]

  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    for (int i = threadIdx.x; i < xn; i += blockDim.x)
      xs[i] = x[(int64_t)b * xn + i];
    for (int i = threadIdx.x; i < gn; i += blockDim.x)
      gys[i] = gy[(int64_t)b * gn + i];
    __syncthreads();

    // weight-gradient partial for this batch element
    for (int i = threadIdx.x; i < wn; i += blockDim.x) {
      const int k = i / (C * R * S_);
      const int c = (i / (R * S_)) % C;
      const int r = (i / S_) % R;
      const int s = i % S_;
      const float* gk = gys + k * OH * OW;
      const float* xc = xs + c * H * W + r * W + s;
      float acc = 0.f;
      for (int oh = 0; oh < OH; ++oh) {
        const float* grow = gk + oh * OW;
        const float* xrow = xc + oh * W;
        for (int ow = 0; ow < OW; ++ow) acc += grow[ow] * xrow[ow];
      }
      atomicAdd(&gw[i], acc);
    }
    if (gb) {
      for (int k = threadIdx.x; k < K; k += blockDim.x) {
        const float* gk = gys + k * OH * OW;
        float acc = 0.f;
        for (int i = 0; i < OH * OW; ++i) acc += gk[i];
        atomicAdd(&gb[k], acc);
      }
    }
    __syncthreads();
  }
  (void)wacc;
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
                                  int B, int K, int N, int fuse_relu) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* ws = smem;  // [N*K]
  const int wn = N * K;
  for (int i = threadIdx.x; i < wn; i += blockDim.x) ws[i] = w[i];
  __syncthreads();

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
                                    int B, int K, int N) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* ws = smem;  // [N*K]
  const int wn = N * K;
  for (int i = threadIdx.x; i < wn; i += blockDim.x) ws[i] = w[i];
  __syncthreads();

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
                                    int B, int K, int N, int b_per_block) {
  const int b0 = blockIdx.x * b_per_block;
  const int b1 = min(B, b0 + b_per_block);
  const int wn = N * K;
  for (int i = threadIdx.x; i < wn; i += blockDim.x) {
    const int n = i / K;
    const int k = i % K;
    float acc = 0.f;
    for (int b = b0; b < b1; ++b) {
      float g = gy[(int64_t)b * N + n];
      if (out && out[(int64_t)b * N + n] <= 0.f) g = 0.f;
      acc += g * x[(int64_t)b * K + k];
    }
    atomicAdd(&gw[i], acc);
  }
  if (gb) {
    for (int n = threadIdx.x; n < N; n += blockDim.x) {
      float acc = 0.f;
      for (int b = b0; b < b1; ++b) {
        float g = gy[(int64_t)b * N + n];
        if (out && out[(int64_t)b * N + n] <= 0.f) g = 0.f;
        acc += g;
      }
      atomicAdd(&gb[n], acc);
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

// ===========================================================================
// host launchers + pybind
// ===========================================================================
namespace {

constexpr int BLK = 256;

void conv2d_fwd(uintptr_t x, uintptr_t w, uintptr_t bias, uintptr_t out,
                int B, int C, int H, int W, int K, int R, int S_,
                uintptr_t stream) {
  const int lds = (C * H * W + K * C * R * S_) * sizeof(float);
  hipLaunchKernelGGL(conv2d_fwd_kernel, dim3(grid_for(B, 1)), dim3(BLK),
                     lds, S(stream), (const float*)x, (const float*)w,
                     (const float*)bias, (float*)out, B, C, H, W, K, R, S_);
}

void conv2d_bwd(uintptr_t x, uintptr_t w, uintptr_t gy, uintptr_t gx,
                uintptr_t gw, uintptr_t gb, int B, int C, int H, int W,
                int K, int R, int S_, uintptr_t stream) {
  const int OH = H - R + 1, OW = W - S_ + 1;
  {
    const int lds = (K * OH * OW + K * C * R * S_) * sizeof(float);
    hipLaunchKernelGGL(conv2d_bwd_x_kernel, dim3(grid_for(B, 1)),
                       dim3(BLK), lds, S(stream), (const float*)gy,
                       (const float*)w, (float*)gx, B, C, H, W, K, R, S_);
  }
  {
    // gw is zeroed by the caller; gb must be zeroed here
    if (gb)
      HIP_CHECK(hipMemsetAsync((void*)gb, 0, K * sizeof(float), S(stream)));
    const int lds =
        (C * H * W + K * OH * OW) * sizeof(float);
    hipLaunchKernelGGL(conv2d_bwd_w_kernel, dim3(grid_for(B, 1)),
                       dim3(BLK), lds, S(stream), (const float*)x,
                       (const float*)gy, (float*)gw, (float*)gb, B, C, H,
                       W, K, R, S_);
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
  const int lds = N * K * sizeof(float);
  hipLaunchKernelGGL(linear_fwd_kernel,
                     dim3(grid_for((int64_t)B * N, BLK)), dim3(BLK), lds,
                     S(stream), (const float*)x, (const float*)w,
                     (const float*)bias, (float*)out, B, K, N,
                     fuse_relu ? 1 : 0);
}

void linear_bwd(uintptr_t x, uintptr_t w, uintptr_t gy, uintptr_t out,
                uintptr_t gx, uintptr_t gw, uintptr_t gb, int B, int K,
                int N, uintptr_t stream) {
  {
    const int lds = N * K * sizeof(float);
    hipLaunchKernelGGL(linear_bwd_x_kernel,
                       dim3(grid_for((int64_t)B * K, BLK)), dim3(BLK), lds,
                       S(stream), (const float*)gy, (const float*)out,
                       (const float*)w, (float*)gx, B, K, N);
  }
  {
    HIP_CHECK(hipMemsetAsync((void*)gw, 0, (size_t)N * K * sizeof(float),
                             S(stream)));
    if (gb)
      HIP_CHECK(hipMemsetAsync((void*)gb, 0, N * sizeof(float), S(stream)));
    const int b_per_block = 16;
    const int blocks = (B + b_per_block - 1) / b_per_block;
    hipLaunchKernelGGL(linear_bwd_w_kernel, dim3(blocks), dim3(BLK), 0,
                       S(stream), (const float*)x, (const float*)gy,
                       (const float*)out, (float*)gw, (float*)gb, B, K, N,
                       b_per_block);
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
  m.def("add_inplace", &add_inplace);
  m.def("reduce_columns", &reduce_columns);
  m.def("scale_f32", &scale_f32);
}
