// 3x3 avg/max pooling fwd/bwd (K9 remainder): the last MIOpen ops on the
// NASNet cell hot path (reference nasnet_utils.py pooling branches; torch
// nn.AvgPool2d(3,s,1,count_include_pad=False) / nn.MaxPool2d(3,s,1)).
//
// NCHW bf16, fp32 accumulation. Max saves the window argmax (u8 0..8) so
// backward is an exact scatter; both backwards are GATHER-style over the
// <=9 covering windows per input position — deterministic, no atomics.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

__global__ __launch_bounds__(256) void pool3_fwd_kernel(
    const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
    unsigned char* __restrict__ argmax, int N, int C, int H, int W, int OH,
    int OW, int stride, int is_max) {
  const int64_t total = (int64_t)N * C * OH * OW;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int ow = (int)(p % OW);
    const int oh = (int)((p / OW) % OH);
    const int64_t nc = p / ((int64_t)OH * OW);
    const bf16_t* xp = x + nc * H * W;
    const int h0 = oh * stride - 1, w0 = ow * stride - 1;
    if (is_max) {
      float best = -3.4e38f;
      int besti = 0;
#pragma unroll
      for (int i = 0; i < 3; ++i) {
#pragma unroll
        for (int j = 0; j < 3; ++j) {
          const int h = h0 + i, w = w0 + j;
          if (h < 0 || h >= H || w < 0 || w >= W) continue;
          const float v = bf2f(xp[h * W + w]);
          if (v > best) {
            best = v;
            besti = i * 3 + j;
          }
        }
      }
      y[p] = f2bf(best);
      if (argmax) argmax[p] = (unsigned char)besti;
    } else {
      float s = 0.f;
      int cnt = 0;
#pragma unroll
      for (int i = 0; i < 3; ++i) {
#pragma unroll
        for (int j = 0; j < 3; ++j) {
          const int h = h0 + i, w = w0 + j;
          if (h < 0 || h >= H || w < 0 || w >= W) continue;
          s += bf2f(xp[h * W + w]);
          ++cnt;
        }
      }
      y[p] = f2bf(s / (float)cnt);  // count_include_pad=False
    }
  }
}

__global__ __launch_bounds__(256) void pool3_bwd_kernel(
    const bf16_t* __restrict__ dy, const unsigned char* __restrict__ argmax,
    bf16_t* __restrict__ dx, int N, int C, int H, int W, int OH, int OW,
    int stride, int is_max) {
  const int64_t total = (int64_t)N * C * H * W;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int w = (int)(p % W);
    const int h = (int)((p / W) % H);
    const int64_t nc = p / ((int64_t)H * W);
    const bf16_t* dyp = dy + nc * OH * OW;
    const unsigned char* am = argmax ? argmax + nc * OH * OW : nullptr;
    float acc = 0.f;
    // covering windows: oh*stride-1 <= h <= oh*stride+1
    for (int oh = max(0, (h - 1 + (stride - 1)) / stride);
         oh <= min(OH - 1, (h + 1) / stride); ++oh) {
      const int i = h - (oh * stride - 1);
      if (i < 0 || i > 2) continue;
      for (int ow = max(0, (w - 1 + (stride - 1)) / stride);
           ow <= min(OW - 1, (w + 1) / stride); ++ow) {
        const int j = w - (ow * stride - 1);
        if (j < 0 || j > 2) continue;
        const int64_t q = (int64_t)oh * OW + ow;
        if (is_max) {
          if (am[q] == (unsigned char)(i * 3 + j)) acc += bf2f(dyp[q]);
        } else {
          const int h0 = oh * stride - 1, w0 = ow * stride - 1;
          const int hc = min(h0 + 2, H - 1) - max(h0, 0) + 1;
          const int wc = min(w0 + 2, W - 1) - max(w0, 0) + 1;
          acc += bf2f(dyp[q]) / (float)(hc * wc);
        }
      }
    }
    dx[p] = f2bf(acc);
  }
}

// Stride-templated backward: the generic kernel's runtime loop bounds
// cost 4 integer divisions per pixel plus a float divide per window —
// measured 150 us/call where the traffic bound is ~15 us
// (profiles/nasprof5_summary.txt). With STRIDE compile-time the <=9
// (stride 1) / <=4 (stride 2) covering windows unroll fully and the
// avg-pool window count becomes a reciprocal-table lookup.
template <int STRIDE, int ISMAX>
__global__ __launch_bounds__(256) void pool3_bwd_tmpl_kernel(
    const bf16_t* __restrict__ dy, const unsigned char* __restrict__ argmax,
    bf16_t* __restrict__ dx, int H, int W, int OH, int OW) {
  const int64_t nc = blockIdx.y;
  const bf16_t* dyp = dy + nc * OH * OW;
  const unsigned char* am = ISMAX ? argmax + nc * OH * OW : nullptr;
  bf16_t* dxp = dx + nc * H * W;
  const int ipx = H * W;
  for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < ipx;
       p += gridDim.x * blockDim.x) {
    const int h = p / W, w = p - (p / W) * W;
    float acc = 0.f;
#pragma unroll
    for (int t = 0; t < (STRIDE == 1 ? 3 : 2); ++t) {
      const int oh = STRIDE == 1 ? h - 1 + t : ((h + 1) >> 1) - t;
      if (oh < 0 || oh >= OH) continue;
      const int i = h - (oh * STRIDE - 1);
      if (STRIDE > 1 && (i < 0 || i > 2)) continue;
#pragma unroll
      for (int u = 0; u < (STRIDE == 1 ? 3 : 2); ++u) {
        const int ow = STRIDE == 1 ? w - 1 + u : ((w + 1) >> 1) - u;
        if (ow < 0 || ow >= OW) continue;
        const int j = w - (ow * STRIDE - 1);
        if (STRIDE > 1 && (j < 0 || j > 2)) continue;
        const int64_t q = (int64_t)oh * OW + ow;
        if (ISMAX) {
          if (am[q] == (unsigned char)(i * 3 + j)) acc += bf2f(dyp[q]);
        } else {
          const int h0 = oh * STRIDE - 1, w0 = ow * STRIDE - 1;
          const int hc = min(h0 + 2, H - 1) - max(h0, 0) + 1;
          const int wc = min(w0 + 2, W - 1) - max(w0, 0) + 1;
          // v_rcp_f32 (~1 ulp) instead of a divide; counts are 1..9 and
          // the result rounds to bf16, so the approximation is exact
          // enough (refchecked against torch fp32).
          acc += bf2f(dyp[q]) * __builtin_amdgcn_rcpf((float)(hc * wc));
        }
      }
    }
    dxp[p] = f2bf(acc);
  }
}

// Per-image forward twin (grid (px_blocks, N*C)): drops the three 64-bit
// divisions per pixel of the flat kernel and uses v_rcp_f32 for the
// avg-pool window count.
template <int ISMAX>
__global__ __launch_bounds__(256) void pool3_fwd_tmpl_kernel(
    const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
    unsigned char* __restrict__ argmax, int H, int W, int OH, int OW,
    int stride) {
  const int64_t nc = blockIdx.y;
  const bf16_t* xp = x + nc * H * W;
  bf16_t* yp = y + nc * OH * OW;
  unsigned char* am = argmax ? argmax + nc * OH * OW : nullptr;
  const int opx = OH * OW;
  for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < opx;
       p += gridDim.x * blockDim.x) {
    const int oh = p / OW, ow = p - (p / OW) * OW;
    const int h0 = oh * stride - 1, w0 = ow * stride - 1;
    if (ISMAX) {
      float best = -3.4e38f;
      int besti = 0;
#pragma unroll
      for (int i = 0; i < 3; ++i) {
#pragma unroll
        for (int j = 0; j < 3; ++j) {
          const int h = h0 + i, w = w0 + j;
          if (h < 0 || h >= H || w < 0 || w >= W) continue;
          const float v = bf2f(xp[h * W + w]);
          if (v > best) {
            best = v;
            besti = i * 3 + j;
          }
        }
      }
      yp[p] = f2bf(best);
      if (am) am[p] = (unsigned char)besti;
    } else {
      float s = 0.f;
      int cnt = 0;
#pragma unroll
      for (int i = 0; i < 3; ++i) {
#pragma unroll
        for (int j = 0; j < 3; ++j) {
          const int h = h0 + i, w = w0 + j;
          if (h < 0 || h >= H || w < 0 || w >= W) continue;
          s += bf2f(xp[h * W + w]);
          ++cnt;
        }
      }
      yp[p] = f2bf(s * __builtin_amdgcn_rcpf((float)cnt));
    }
  }
}

int pgrid(int64_t total) {
  return (int)std::min<int64_t>((total + 255) / 256, 2048);
}

}  // namespace

void pool3_fwd(const at::Tensor& x, at::Tensor& y,
               const c10::optional<at::Tensor>& argmax, int64_t stride,
               int64_t is_max) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous() &&
              x.scalar_type() == at::kBFloat16, "pool3: bf16 NCHW");
  const int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3);
  const int OH = (int)y.size(2), OW = (int)y.size(3);
  auto stream = at::cuda::getCurrentCUDAStream();
  unsigned char* am = nullptr;
  if (argmax.has_value() && argmax->defined())
    am = (unsigned char*)argmax->data_ptr();
  const int64_t bc = (int64_t)N * C;
  if (bc <= 65535) {
    const int px_blocks =
        std::max(1, std::min((OH * OW + 255) / 256, (int)(2048 / bc) + 1));
    auto kern = is_max ? pool3_fwd_tmpl_kernel<1> : pool3_fwd_tmpl_kernel<0>;
    hipLaunchKernelGGL(kern, dim3((unsigned)px_blocks, (unsigned)bc),
                       dim3(256), 0, stream.stream(),
                       (const bf16_t*)x.data_ptr(), (bf16_t*)y.data_ptr(),
                       am, H, W, OH, OW, (int)stride);
    HIP_CHECK_KERNEL();
    return;
  }
  hipLaunchKernelGGL(pool3_fwd_kernel, dim3(pgrid(y.numel())), dim3(256), 0,
                     stream.stream(), (const bf16_t*)x.data_ptr(),
                     (bf16_t*)y.data_ptr(), am, N, C, H, W, OH, OW,
                     (int)stride, (int)is_max);
  HIP_CHECK_KERNEL();
}

void pool3_bwd(const at::Tensor& dy, const c10::optional<at::Tensor>& argmax,
               at::Tensor& dx, int64_t stride, int64_t is_max) {
  const int N = (int)dx.size(0), C = (int)dx.size(1), H = (int)dx.size(2),
            W = (int)dx.size(3);
  const int OH = (int)dy.size(2), OW = (int)dy.size(3);
  auto stream = at::cuda::getCurrentCUDAStream();
  const unsigned char* am = nullptr;
  if (argmax.has_value() && argmax->defined())
    am = (const unsigned char*)argmax->data_ptr();
  TORCH_CHECK(!is_max || am, "pool3_bwd: max pooling needs saved argmax");
  const int64_t bc = (int64_t)N * C;
  if ((stride == 1 || stride == 2) && bc <= 65535) {
    const int px_blocks =
        std::max(1, std::min((H * W + 255) / 256, (int)(2048 / bc) + 1));
    using kern_t = void (*)(const bf16_t*, const unsigned char*, bf16_t*,
                            int, int, int, int);
    kern_t kern;
    if (stride == 1)
      kern = is_max ? pool3_bwd_tmpl_kernel<1, 1> : pool3_bwd_tmpl_kernel<1, 0>;
    else
      kern = is_max ? pool3_bwd_tmpl_kernel<2, 1> : pool3_bwd_tmpl_kernel<2, 0>;
    hipLaunchKernelGGL(kern, dim3((unsigned)px_blocks, (unsigned)bc),
                       dim3(256), 0, stream.stream(),
                       (const bf16_t*)dy.data_ptr(), am,
                       (bf16_t*)dx.data_ptr(), H, W, OH, OW);
    HIP_CHECK_KERNEL();
    return;
  }
  hipLaunchKernelGGL(pool3_bwd_kernel, dim3(pgrid(dx.numel())), dim3(256), 0,
                     stream.stream(), (const bf16_t*)dy.data_ptr(), am,
                     (bf16_t*)dx.data_ptr(), N, C, H, W, OH, OW, (int)stride,
                     (int)is_max);
  HIP_CHECK_KERNEL();
}
