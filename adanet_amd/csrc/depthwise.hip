// Depthwise 2-D convolution (K9): the other half of the NASNet separable
// conv (reference nasnet_utils.py:182 _stacked_separable_conv = depthwise
// KxK then pointwise 1x1; the pointwise half runs on gemm_tr_batched).
// Depthwise conv has no reuse across channels, so it is bandwidth-bound
// elementwise work: one thread per output pixel, serial tap loop (9/25
// taps), coalesced along the contiguous W dimension. Works for any channel
// count (no GEMM alignment constraints) at stride 1 or 2.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

__global__ __launch_bounds__(256) void depthwise_fwd_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
    bf16_t* __restrict__ y, int B, int C, int H, int W, int OH, int OW,
    int KS, int stride, int pad) {
  const int64_t total = (int64_t)B * C * OH * OW;
  const int64_t gstride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += gstride) {
    const int ow = (int)(p % OW);
    const int oh = (int)((p / OW) % OH);
    const int c = (int)((p / ((int64_t)OW * OH)) % C);
    const int b = (int)(p / ((int64_t)OW * OH * C));
    const bf16_t* xp = x + ((int64_t)b * C + c) * H * W;
    const bf16_t* wp = w + c * KS * KS;
    float acc = 0.f;
    for (int kh = 0; kh < KS; ++kh) {
      const int ih = oh * stride + kh - pad;
      if (ih < 0 || ih >= H) continue;
      for (int kw = 0; kw < KS; ++kw) {
        const int iw = ow * stride + kw - pad;
        if (iw < 0 || iw >= W) continue;
        acc += bf2f(wp[kh * KS + kw]) * bf2f(xp[ih * W + iw]);
      }
    }
    y[p] = f2bf(acc);
  }
}

// KS-templated fast path: one (batch, channel) image per blockIdx.y, so
// the KSxKS tap weights load once into registers (same c for the whole
// block -> broadcast from L1) and the tap loops fully unroll.
template <int KS>
__global__ __launch_bounds__(256) void depthwise_fwd_tmpl_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
    bf16_t* __restrict__ y, int C, int H, int W, int OH, int OW, int stride,
    int pad) {
  const int bc = blockIdx.y;
  const int c = bc % C;
  const bf16_t* xp = x + (int64_t)bc * H * W;
  bf16_t* yp = y + (int64_t)bc * OH * OW;
  float wr[KS * KS];
#pragma unroll
  for (int i = 0; i < KS * KS; ++i) wr[i] = bf2f(w[c * KS * KS + i]);
  const int total = OH * OW;
  for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += gridDim.x * blockDim.x) {
    const int ow = p % OW, oh = p / OW;
    float acc = 0.f;
#pragma unroll
    for (int kh = 0; kh < KS; ++kh) {
      const int ih = oh * stride + kh - pad;
      if (ih < 0 || ih >= H) continue;
#pragma unroll
      for (int kw = 0; kw < KS; ++kw) {
        const int iw = ow * stride + kw - pad;
        if (iw < 0 || iw >= W) continue;
        acc += wr[kh * KS + kw] * bf2f(xp[ih * W + iw]);
      }
    }
    yp[p] = f2bf(acc);
  }
}

// LDS-staged forward: the tap loop re-reads each input pixel KS^2 times;
// from L1 that measured ~25x off the bandwidth roofline (fwd<5> 141 us
// where x+y once is ~9 us, profiles/nasprof5_summary.txt). Stage a GROUP
// of P whole (b,c) images in LDS (images are contiguous in NCHW, so the
// group load is one coalesced span), then compute every output pixel
// from LDS. Weights stay in L1 (C*KS^2*2B is a few KB, fully resident).
template <int KS>
__global__ __launch_bounds__(256) void depthwise_fwd_lds_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
    bf16_t* __restrict__ y, int C, int H, int W, int OH, int OW, int stride,
    int pad, int P, int BC) {
  __shared__ bf16_t xs[8192];  // 16 KB: P * H * W <= 8192 by host contract
  const int g0 = blockIdx.y * P;
  const int ni = min(P, BC - g0);
  const int npx_in = ni * H * W;
  const bf16_t* gx = x + (int64_t)g0 * H * W;
  if ((npx_in & 7) == 0) {  // image sizes are multiples of 8 in practice
    const uint4* src = (const uint4*)gx;
    uint4* dst = (uint4*)xs;
    for (int i = threadIdx.x; i < (npx_in >> 3); i += 256) dst[i] = src[i];
  } else {
    for (int i = threadIdx.x; i < npx_in; i += 256) xs[i] = gx[i];
  }
  __syncthreads();
  const int opx = OH * OW;
  bf16_t* gy = y + (int64_t)g0 * opx;
  if (opx >= 256) {
    // image-outer: whole block works one image at a time so the KS^2 tap
    // weights live in registers (the flat loop re-read them from L1 per
    // pixel — measured slower than the old register-tap kernel).
    for (int img = 0; img < ni; ++img) {
      const int c = (g0 + img) % C;
      float wr[KS * KS];
#pragma unroll
      for (int i = 0; i < KS * KS; ++i) wr[i] = bf2f(w[c * KS * KS + i]);
      const bf16_t* xi = xs + img * H * W;
      bf16_t* yi = gy + (int64_t)img * opx;
      for (int p = threadIdx.x; p < opx; p += 256) {
        const int oh = p / OW, ow = p - (p / OW) * OW;
        const int ih0 = oh * stride - pad, iw0 = ow * stride - pad;
        float acc = 0.f;
#pragma unroll
        for (int kh = 0; kh < KS; ++kh) {
          const int ih = ih0 + kh;
          if (ih < 0 || ih >= H) continue;
#pragma unroll
          for (int kw = 0; kw < KS; ++kw) {
            const int iw = iw0 + kw;
            if (iw < 0 || iw >= W) continue;
            acc += wr[kh * KS + kw] * bf2f(xi[ih * W + iw]);
          }
        }
        yi[p] = f2bf(acc);
      }
    }
    return;
  }
  const int npx_out = ni * opx;
  for (int p = threadIdx.x; p < npx_out; p += 256) {
    const int img = p / opx;
    const int rem = p - img * opx;
    const int oh = rem / OW, ow = rem - (rem / OW) * OW;
    const int c = (g0 + img) % C;
    const bf16_t* wp = w + c * KS * KS;
    const bf16_t* xi = xs + img * H * W;
    const int ih0 = oh * stride - pad, iw0 = ow * stride - pad;
    float acc = 0.f;
#pragma unroll
    for (int kh = 0; kh < KS; ++kh) {
      const int ih = ih0 + kh;
      if (ih < 0 || ih >= H) continue;
#pragma unroll
      for (int kw = 0; kw < KS; ++kw) {
        const int iw = iw0 + kw;
        if (iw < 0 || iw >= W) continue;
        acc += bf2f(wp[kh * KS + kw]) * bf2f(xi[ih * W + iw]);
      }
    }
    gy[p] = f2bf(acc);
  }
}

// LDS-staged dX: same grouping, dy staged in LDS, full-correlation taps.
template <int KS, int STRIDE>
__global__ __launch_bounds__(256) void depthwise_bwd_dx_lds_kernel(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ w,
    bf16_t* __restrict__ dx, int C, int H, int W, int OH, int OW, int pad,
    int P, int BC) {
  __shared__ bf16_t ds[8192];  // P * OH * OW <= 8192 by host contract
  const int g0 = blockIdx.y * P;
  const int ni = min(P, BC - g0);
  const int opx = OH * OW;
  const int nin = ni * opx;
  const bf16_t* gdy = dy + (int64_t)g0 * opx;
  if ((nin & 7) == 0) {
    const uint4* src = (const uint4*)gdy;
    uint4* dst = (uint4*)ds;
    for (int i = threadIdx.x; i < (nin >> 3); i += 256) dst[i] = src[i];
  } else {
    for (int i = threadIdx.x; i < nin; i += 256) ds[i] = gdy[i];
  }
  __syncthreads();
  const int ipx = H * W;
  bf16_t* gdx = dx + (int64_t)g0 * ipx;
  if (ipx >= 256) {
    // image-outer: tap weights in registers for the whole image.
    for (int img = 0; img < ni; ++img) {
      const int c = (g0 + img) % C;
      float wr[KS * KS];
#pragma unroll
      for (int i = 0; i < KS * KS; ++i) wr[i] = bf2f(w[c * KS * KS + i]);
      const bf16_t* di = ds + img * opx;
      bf16_t* xo = gdx + (int64_t)img * ipx;
      for (int p = threadIdx.x; p < ipx; p += 256) {
        const int ih = p / W, iw = p - (p / W) * W;
        float acc = 0.f;
#pragma unroll
        for (int kh = 0; kh < KS; ++kh) {
          const int num_h = ih + pad - kh;
          if (num_h < 0 || (STRIDE > 1 && (num_h % STRIDE))) continue;
          const int oh = num_h / STRIDE;
          if (oh >= OH) continue;
#pragma unroll
          for (int kw = 0; kw < KS; ++kw) {
            const int num_w = iw + pad - kw;
            if (num_w < 0 || (STRIDE > 1 && (num_w % STRIDE))) continue;
            const int ow = num_w / STRIDE;
            if (ow >= OW) continue;
            acc += wr[kh * KS + kw] * bf2f(di[oh * OW + ow]);
          }
        }
        xo[p] = f2bf(acc);
      }
    }
    return;
  }
  // small images: flat loop over every output pixel of the group (weights
  // from L1 — C is large exactly when images are small).
  for (int p = threadIdx.x; p < ni * ipx; p += 256) {
    const int img = p / ipx;
    const int rem = p - img * ipx;
    const int ih = rem / W, iw = rem - (rem / W) * W;
    const int c = (g0 + img) % C;
    const bf16_t* wp = w + c * KS * KS;
    const bf16_t* di = ds + img * opx;
    float acc = 0.f;
#pragma unroll
    for (int kh = 0; kh < KS; ++kh) {
      const int num_h = ih + pad - kh;
      if (num_h < 0 || (STRIDE > 1 && (num_h % STRIDE))) continue;
      const int oh = num_h / STRIDE;
      if (oh >= OH) continue;
#pragma unroll
      for (int kw = 0; kw < KS; ++kw) {
        const int num_w = iw + pad - kw;
        if (num_w < 0 || (STRIDE > 1 && (num_w % STRIDE))) continue;
        const int ow = num_w / STRIDE;
        if (ow >= OW) continue;
        acc += bf2f(wp[kh * KS + kw]) * bf2f(di[oh * OW + ow]);
      }
    }
    gdx[p] = f2bf(acc);
  }
}

// LDS-staged dW: grid (C, nchunks); each block owns a span of images b for
// ONE channel, stages x[b,c] and dy[b,c] image groups in LDS, accumulates
// all KS^2 taps in registers from LDS, and writes its per-chunk partial
// into a workspace slot; depthwise_dw_reduce_kernel sums chunks in fixed
// order (deterministic — replaces the per-tap atomicAdd, whose fp32
// ordering varied run to run).
template <int KS>
__global__ __launch_bounds__(256) void depthwise_bwd_dw_lds_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ dy,
    float* __restrict__ wsp, int B, int C, int H, int W, int OH, int OW,
    int stride, int pad, int b_per_chunk, int G) {
  __shared__ bf16_t xs[4096];  // G * H * W <= 4096 by host contract
  __shared__ bf16_t ds[4096];  // G * OH * OW <= 4096
  const int c = blockIdx.x;
  const int b0 = blockIdx.y * b_per_chunk;
  const int b1 = min(B, b0 + b_per_chunk);
  const int ipx = H * W, opx = OH * OW;
  float acc[KS * KS];
#pragma unroll
  for (int t = 0; t < KS * KS; ++t) acc[t] = 0.f;
  for (int g = b0; g < b1; g += G) {
    const int ni = min(G, b1 - g);
    if ((ipx & 7) == 0) {
      for (int i = threadIdx.x; i < ((ni * ipx) >> 3); i += 256) {
        const int img = i / (ipx >> 3);
        ((uint4*)xs)[i] = ((const uint4*)(
            x + ((int64_t)(g + img) * C + c) * ipx))[i - img * (ipx >> 3)];
      }
    } else {
      for (int i = threadIdx.x; i < ni * ipx; i += 256) {
        const int img = i / ipx;
        xs[i] = x[((int64_t)(g + img) * C + c) * ipx + (i - img * ipx)];
      }
    }
    if ((opx & 7) == 0) {
      for (int i = threadIdx.x; i < ((ni * opx) >> 3); i += 256) {
        const int img = i / (opx >> 3);
        ((uint4*)ds)[i] = ((const uint4*)(
            dy + ((int64_t)(g + img) * C + c) * opx))[i - img * (opx >> 3)];
      }
    } else {
      for (int i = threadIdx.x; i < ni * opx; i += 256) {
        const int img = i / opx;
        ds[i] = dy[((int64_t)(g + img) * C + c) * opx + (i - img * opx)];
      }
    }
    __syncthreads();
    for (int p = threadIdx.x; p < ni * opx; p += 256) {
      const int img = p / opx;
      const int rem = p - img * opx;
      const int oh = rem / OW, ow = rem - (rem / OW) * OW;
      const float dyv = bf2f(ds[p]);
      const bf16_t* xi = xs + img * ipx;
      const int ih0 = oh * stride - pad, iw0 = ow * stride - pad;
#pragma unroll
      for (int kh = 0; kh < KS; ++kh) {
        const int ih = ih0 + kh;
        if (ih < 0 || ih >= H) continue;
#pragma unroll
        for (int kw = 0; kw < KS; ++kw) {
          const int iw = iw0 + kw;
          if (iw < 0 || iw >= W) continue;
          acc[kh * KS + kw] += dyv * bf2f(xi[ih * W + iw]);
        }
      }
    }
    __syncthreads();
  }
  __shared__ float partial[4];
#pragma unroll
  for (int t = 0; t < KS * KS; ++t) {
    float v = wave_reduce_sum(acc[t]);
    if ((threadIdx.x & 63) == 0) partial[threadIdx.x >> 6] = v;
    __syncthreads();
    if (threadIdx.x == 0) {
      wsp[((int64_t)blockIdx.y * C + c) * (KS * KS) + t] =
          partial[0] + partial[1] + partial[2] + partial[3];
    }
    __syncthreads();
  }
}

// dw[c*KSQ+t] = sum over chunks of wsp[(chunk*C + c)*KSQ + t], fixed order.
__global__ __launch_bounds__(256) void depthwise_dw_reduce_kernel(
    const float* __restrict__ wsp, float* __restrict__ dw, int C, int KSQ,
    int nchunks) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= C * KSQ) return;
  float s = 0.f;
  for (int r = 0; r < nchunks; ++r) s += wsp[(int64_t)r * C * KSQ + i];
  dw[i] = s;
}

// dX templated fast path (KS, STRIDE compile-time): one (batch, channel)
// image per blockIdx.y so the taps load once into registers and the loops
// fully unroll — the runtime-KS kernel below was 13.6% of the improve_nas
// step at 0.3 TB/s (branchy runtime tap loop, per-element weight reloads).
template <int KS, int STRIDE>
__global__ __launch_bounds__(256) void depthwise_bwd_dx_tmpl_kernel(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ w,
    bf16_t* __restrict__ dx, int C, int H, int W, int OH, int OW, int pad) {
  const int bc = blockIdx.y;
  const int c = bc % C;
  const bf16_t* dyp = dy + (int64_t)bc * OH * OW;
  bf16_t* dxp = dx + (int64_t)bc * H * W;
  float wr[KS * KS];
#pragma unroll
  for (int i = 0; i < KS * KS; ++i) wr[i] = bf2f(w[c * KS * KS + i]);
  const int total = H * W;
  for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += gridDim.x * blockDim.x) {
    const int iw = p % W;
    const int ih = p / W;
    float acc = 0.f;
#pragma unroll
    for (int kh = 0; kh < KS; ++kh) {
      const int num_h = ih + pad - kh;
      if (num_h < 0 || (STRIDE > 1 && (num_h % STRIDE))) continue;
      const int oh = num_h / STRIDE;
      if (oh >= OH) continue;
#pragma unroll
      for (int kw = 0; kw < KS; ++kw) {
        const int num_w = iw + pad - kw;
        if (num_w < 0 || (STRIDE > 1 && (num_w % STRIDE))) continue;
        const int ow = num_w / STRIDE;
        if (ow >= OW) continue;
        acc += wr[kh * KS + kw] * bf2f(dyp[oh * OW + ow]);
      }
    }
    dxp[p] = f2bf(acc);
  }
}

// dX: full correlation with the flipped kernel, honoring stride divisibility.
__global__ __launch_bounds__(256) void depthwise_bwd_dx_kernel(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ w,
    bf16_t* __restrict__ dx, int B, int C, int H, int W, int OH, int OW,
    int KS, int stride, int pad) {
  const int64_t total = (int64_t)B * C * H * W;
  const int64_t gstride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += gstride) {
    const int iw = (int)(p % W);
    const int ih = (int)((p / W) % H);
    const int c = (int)((p / ((int64_t)W * H)) % C);
    const int b = (int)(p / ((int64_t)W * H * C));
    const bf16_t* dyp = dy + ((int64_t)b * C + c) * OH * OW;
    const bf16_t* wp = w + c * KS * KS;
    float acc = 0.f;
    for (int kh = 0; kh < KS; ++kh) {
      const int num_h = ih + pad - kh;
      if (num_h < 0 || num_h % stride) continue;
      const int oh = num_h / stride;
      if (oh >= OH) continue;
      for (int kw = 0; kw < KS; ++kw) {
        const int num_w = iw + pad - kw;
        if (num_w < 0 || num_w % stride) continue;
        const int ow = num_w / stride;
        if (ow >= OW) continue;
        acc += bf2f(wp[kh * KS + kw]) * bf2f(dyp[oh * OW + ow]);
      }
    }
    dx[p] = f2bf(acc);
  }
}

// dW: one block per (channel, tap); block-parallel reduction over b,oh,ow.
// Single-pass dW: each thread reads dy ONCE per output point and
// accumulates all KS*KS taps in registers (neighboring x reads hit L2),
// instead of the KS^2 full re-reads of the tap-per-block kernel
// (measured 14.5% of the improve_nas step at 0.7 TB/s,
// profiles/nasprof_summary.txt). grid = (C, chunks); one atomicAdd per
// (block, tap).
template <int KS>
__global__ __launch_bounds__(256) void depthwise_bwd_dw_fused_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ dy,
    float* __restrict__ dw, int B, int C, int H, int W, int OH, int OW,
    int stride, int pad, int64_t chunk) {
  const int c = blockIdx.x;
  const int64_t total = (int64_t)B * OH * OW;
  const int64_t start = (int64_t)blockIdx.y * chunk;
  const int64_t end = min(total, start + chunk);
  float acc[KS * KS];
#pragma unroll
  for (int t = 0; t < KS * KS; ++t) acc[t] = 0.f;
  for (int64_t p = start + threadIdx.x; p < end; p += blockDim.x) {
    const int ow = (int)(p % OW);
    const int oh = (int)((p / OW) % OH);
    const int b = (int)(p / ((int64_t)OW * OH));
    const float dyv =
        bf2f(dy[((int64_t)b * C + c) * OH * OW + oh * OW + ow]);
    const bf16_t* xp = x + ((int64_t)b * C + c) * H * W;
    const int ih0 = oh * stride - pad, iw0 = ow * stride - pad;
#pragma unroll
    for (int kh = 0; kh < KS; ++kh) {
      const int ih = ih0 + kh;
      if (ih < 0 || ih >= H) continue;
#pragma unroll
      for (int kw = 0; kw < KS; ++kw) {
        const int iw = iw0 + kw;
        if (iw < 0 || iw >= W) continue;
        acc[kh * KS + kw] += dyv * bf2f(xp[ih * W + iw]);
      }
    }
  }
  __shared__ float partial[4];
#pragma unroll
  for (int t = 0; t < KS * KS; ++t) {
    float v = wave_reduce_sum(acc[t]);
    if ((threadIdx.x & 63) == 0) partial[threadIdx.x >> 6] = v;
    __syncthreads();
    if (threadIdx.x == 0) {
      atomicAdd(&dw[c * KS * KS + t],
                partial[0] + partial[1] + partial[2] + partial[3]);
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(256) void depthwise_bwd_dw_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ dy,
    float* __restrict__ dw, int B, int C, int H, int W, int OH, int OW,
    int KS, int stride, int pad) {
  const int c = blockIdx.x;
  const int kh = blockIdx.y / KS, kw = blockIdx.y % KS;
  const int64_t total = (int64_t)B * OH * OW;
  float acc = 0.f;
  for (int64_t p = threadIdx.x; p < total; p += blockDim.x) {
    const int ow = (int)(p % OW);
    const int oh = (int)((p / OW) % OH);
    const int b = (int)(p / ((int64_t)OW * OH));
    const int ih = oh * stride + kh - pad;
    const int iw = ow * stride + kw - pad;
    if (ih < 0 || ih >= H || iw < 0 || iw >= W) continue;
    acc += bf2f(dy[((int64_t)b * C + c) * OH * OW + oh * OW + ow]) *
           bf2f(x[((int64_t)b * C + c) * H * W + ih * W + iw]);
  }
  acc = wave_reduce_sum(acc);
  __shared__ float partial[4];
  if ((threadIdx.x & 63) == 0) partial[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    dw[c * KS * KS + kh * KS + kw] =
        partial[0] + partial[1] + partial[2] + partial[3];
  }
}

static int dw_grid(int64_t n) {
  return (int)std::min<int64_t>((n + 255) / 256, 4096);
}

static bool dw_old_path() {
  static const bool v = getenv("ADANET_DW_OLD") != nullptr;
  return v;
}

void depthwise_fwd(const at::Tensor& x, const at::Tensor& w, at::Tensor& y,
                   int64_t stride, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 &&
                  x.is_contiguous() && w.is_contiguous(),
              "depthwise: contiguous bf16");
  const int B = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3);
  const int OH = (int)y.size(2), OW = (int)y.size(3);
  const int KS = (int)w.size(w.dim() - 1);
  auto stream = at::cuda::getCurrentCUDAStream();
  const int64_t bc = (int64_t)B * C;
  if ((KS == 3 || KS == 5 || KS == 7) && H * W <= 8192 && !dw_old_path()) {
    // P bounded by LDS (staged input) AND by output work per block (keep
    // blocks small enough that ceil(BC/P) fills the chip).
    const int P = std::max(1, std::min((int)(8192 / (H * W)),
                                       std::max(1, 4096 / (OH * OW))));
    const int groups = (int)((bc + P - 1) / P);
    auto kern = KS == 3   ? depthwise_fwd_lds_kernel<3>
                : KS == 5 ? depthwise_fwd_lds_kernel<5>
                          : depthwise_fwd_lds_kernel<7>;
    hipLaunchKernelGGL(kern, dim3(1, (unsigned)groups), dim3(256), 0,
                       stream.stream(), (const bf16_t*)x.data_ptr(),
                       (const bf16_t*)w.data_ptr(), (bf16_t*)y.data_ptr(), C,
                       H, W, OH, OW, (int)stride, (int)pad, P, (int)bc);
  } else if ((KS == 3 || KS == 5) && bc <= 65535) {
    const int px_blocks = std::max(1, std::min((OH * OW + 255) / 256,
                                               (int)(2048 / bc) + 1));
    auto kern = KS == 3 ? depthwise_fwd_tmpl_kernel<3>
                        : depthwise_fwd_tmpl_kernel<5>;
    hipLaunchKernelGGL(kern, dim3((unsigned)px_blocks, (unsigned)bc),
                       dim3(256), 0, stream.stream(),
                       (const bf16_t*)x.data_ptr(),
                       (const bf16_t*)w.data_ptr(), (bf16_t*)y.data_ptr(), C,
                       H, W, OH, OW, (int)stride, (int)pad);
  } else {
    hipLaunchKernelGGL(depthwise_fwd_kernel,
                       dim3(dw_grid((int64_t)B * C * OH * OW)), dim3(256), 0,
                       stream.stream(), (const bf16_t*)x.data_ptr(),
                       (const bf16_t*)w.data_ptr(), (bf16_t*)y.data_ptr(), B,
                       C, H, W, OH, OW, KS, (int)stride, (int)pad);
  }
  HIP_CHECK_KERNEL();
}

void depthwise_bwd_dx(const at::Tensor& dy, const at::Tensor& w,
                      at::Tensor& dx, int64_t stride, int64_t pad) {
  const int B = (int)dx.size(0), C = (int)dx.size(1), H = (int)dx.size(2),
            W = (int)dx.size(3);
  const int OH = (int)dy.size(2), OW = (int)dy.size(3);
  const int KS = (int)w.size(w.dim() - 1);
  auto stream = at::cuda::getCurrentCUDAStream();
  const int64_t bc = (int64_t)B * C;
  if ((KS == 3 || KS == 5 || KS == 7) && (stride == 1 || stride == 2) &&
      OH * OW <= 8192 && !dw_old_path()) {
    const int P = std::max(1, std::min((int)(8192 / (OH * OW)),
                                       std::max(1, 4096 / (H * W))));
    const int groups = (int)((bc + P - 1) / P);
    using kern_t = void (*)(const bf16_t*, const bf16_t*, bf16_t*, int, int,
                            int, int, int, int, int, int);
    kern_t kern;
    if (stride == 1) {
      kern = KS == 3   ? depthwise_bwd_dx_lds_kernel<3, 1>
             : KS == 5 ? depthwise_bwd_dx_lds_kernel<5, 1>
                       : depthwise_bwd_dx_lds_kernel<7, 1>;
    } else {
      kern = KS == 3   ? depthwise_bwd_dx_lds_kernel<3, 2>
             : KS == 5 ? depthwise_bwd_dx_lds_kernel<5, 2>
                       : depthwise_bwd_dx_lds_kernel<7, 2>;
    }
    hipLaunchKernelGGL(kern, dim3(1, (unsigned)groups), dim3(256), 0,
                       stream.stream(), (const bf16_t*)dy.data_ptr(),
                       (const bf16_t*)w.data_ptr(), (bf16_t*)dx.data_ptr(),
                       C, H, W, OH, OW, (int)pad, P, (int)bc);
    HIP_CHECK_KERNEL();
    return;
  }
  if ((KS == 3 || KS == 5 || KS == 7) && (stride == 1 || stride == 2) &&
      bc <= 65535) {
    const int px_blocks = std::max(1, std::min((H * W + 255) / 256,
                                               (int)(2048 / bc) + 1));
    using kern_t = void (*)(const bf16_t*, const bf16_t*, bf16_t*, int, int,
                            int, int, int, int);
    kern_t kern;
    if (stride == 1) {
      kern = KS == 3 ? depthwise_bwd_dx_tmpl_kernel<3, 1>
             : KS == 5 ? depthwise_bwd_dx_tmpl_kernel<5, 1>
                       : depthwise_bwd_dx_tmpl_kernel<7, 1>;
    } else {
      kern = KS == 3 ? depthwise_bwd_dx_tmpl_kernel<3, 2>
             : KS == 5 ? depthwise_bwd_dx_tmpl_kernel<5, 2>
                       : depthwise_bwd_dx_tmpl_kernel<7, 2>;
    }
    hipLaunchKernelGGL(kern, dim3((unsigned)px_blocks, (unsigned)bc),
                       dim3(256), 0, stream.stream(),
                       (const bf16_t*)dy.data_ptr(),
                       (const bf16_t*)w.data_ptr(), (bf16_t*)dx.data_ptr(),
                       C, H, W, OH, OW, (int)pad);
    HIP_CHECK_KERNEL();
    return;
  }
  hipLaunchKernelGGL(depthwise_bwd_dx_kernel,
                     dim3(dw_grid((int64_t)B * C * H * W)), dim3(256), 0,
                     stream.stream(), (const bf16_t*)dy.data_ptr(),
                     (const bf16_t*)w.data_ptr(), (bf16_t*)dx.data_ptr(), B,
                     C, H, W, OH, OW, KS, (int)stride, (int)pad);
  HIP_CHECK_KERNEL();
}

void depthwise_bwd_dw(const at::Tensor& x, const at::Tensor& dy,
                      at::Tensor& dw, int64_t stride, int64_t pad) {
  const int B = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3);
  const int OH = (int)dy.size(2), OW = (int)dy.size(3);
  const int KS = (int)(std::lround(std::sqrt((double)(dw.numel() / C))));
  TORCH_CHECK(dw.scalar_type() == at::kFloat, "depthwise dW: fp32 out");
  auto stream = at::cuda::getCurrentCUDAStream();
  const int64_t total = (int64_t)B * OH * OW;
  if ((KS == 3 || KS == 5 || KS == 7) && H * W <= 4096 && OH * OW <= 4096 &&
      !dw_old_path()) {
    // LDS-staged deterministic path: per-chunk workspace + ordered reduce.
    const int nchunks =
        (int)std::max<int64_t>(1, std::min<int64_t>(1024 / C, B));
    const int b_per_chunk = (B + nchunks - 1) / nchunks;
    const int G = std::max(
        1, std::min({(int)(4096 / (H * W)), (int)(4096 / (OH * OW)), B}));
    auto wsp = at::empty({nchunks, C, KS * KS}, dw.options());
#define LAUNCH_DWL(KSV)                                                           hipLaunchKernelGGL((depthwise_bwd_dw_lds_kernel<KSV>),                                           dim3((unsigned)C, (unsigned)nchunks), dim3(256), 0,                           stream.stream(), (const bf16_t*)x.data_ptr(),                                 (const bf16_t*)dy.data_ptr(), wsp.data_ptr<float>(),                          B, C, H, W, OH, OW, (int)stride, (int)pad,                                    b_per_chunk, G)
    if (KS == 3) LAUNCH_DWL(3);
    else if (KS == 5) LAUNCH_DWL(5);
    else LAUNCH_DWL(7);
#undef LAUNCH_DWL
    hipLaunchKernelGGL(depthwise_dw_reduce_kernel,
                       dim3((unsigned)((C * KS * KS + 255) / 256)), dim3(256),
                       0, stream.stream(), wsp.data_ptr<float>(),
                       dw.data_ptr<float>(), C, KS * KS, nchunks);
    HIP_CHECK_KERNEL();
    return;
  }
  // single-pass fused kernel for the NASNet tap sizes; dw must be zeroed
  // by the caller contract (it is freshly allocated in ops/conv.py).
  if (KS == 3 || KS == 5 || KS == 7) {
    const int nchunks =
        (int)std::max<int64_t>(1, std::min<int64_t>(1024 / C, 64));
    const int64_t chunk = (total + nchunks - 1) / nchunks;
    dw.zero_();
#define LAUNCH_DWF(KSV)                                                       \
    hipLaunchKernelGGL((depthwise_bwd_dw_fused_kernel<KSV>),                  \
                       dim3((unsigned)C, (unsigned)nchunks), dim3(256), 0,    \
                       stream.stream(), (const bf16_t*)x.data_ptr(),          \
                       (const bf16_t*)dy.data_ptr(), dw.data_ptr<float>(),    \
                       B, C, H, W, OH, OW, (int)stride, (int)pad, chunk)
    if (KS == 3) LAUNCH_DWF(3);
    else if (KS == 5) LAUNCH_DWF(5);
    else LAUNCH_DWF(7);
#undef LAUNCH_DWF
    HIP_CHECK_KERNEL();
    return;
  }
  hipLaunchKernelGGL(depthwise_bwd_dw_kernel,
                     dim3((unsigned)C, (unsigned)(KS * KS)), dim3(256), 0,
                     stream.stream(), (const bf16_t*)x.data_ptr(),
                     (const bf16_t*)dy.data_ptr(), dw.data_ptr<float>(), B, C,
                     H, W, OH, OW, KS, (int)stride, (int)pad);
  HIP_CHECK_KERNEL();
}
