// Batched im2col / col2im for the unfold-GEMM conv path (K9 tail).
//
// torch's F.unfold launches at::native::im2col once PER IMAGE (138k
// launches of 3.7µs in one improve_nas iteration — pure launch overhead,
// profiles/nasprof2_summary.txt); these kernels do the whole batch in one
// grid-stride launch and fold the K-dim zero-padding to 32 (the MFMA GEMM
// alignment) into the same pass, removing the separate F.pad kernel too.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

__global__ __launch_bounds__(256) void im2col_bf16_kernel(
    const bf16_t* __restrict__ x, bf16_t* __restrict__ out, int B, int C,
    int H, int W, int K, int OH, int OW, int stride, int pad, int ckk_pad) {
  const int64_t L = (int64_t)OH * OW;
  const int64_t total = (int64_t)B * ckk_pad * L;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int64_t l = p % L;
    const int row = (int)((p / L) % ckk_pad);
    const int b = (int)(p / (L * ckk_pad));
    if (row >= C * K * K) {  // GEMM-alignment zero pad
      out[p] = f2bf(0.f);
      continue;
    }
    const int c = row / (K * K);
    const int kh = (row / K) % K;
    const int kw = row % K;
    const int oh = (int)(l / OW), ow = (int)(l % OW);
    const int ih = oh * stride + kh - pad;
    const int iw = ow * stride + kw - pad;
    out[p] = (ih >= 0 && ih < H && iw >= 0 && iw < W)
                 ? x[(((int64_t)b * C + c) * H + ih) * W + iw]
                 : f2bf(0.f);
  }
}

__global__ __launch_bounds__(256) void col2im_bf16_kernel(
    const bf16_t* __restrict__ du, bf16_t* __restrict__ dx, int B, int C,
    int H, int W, int K, int OH, int OW, int stride, int pad, int ckk_pad) {
  const int64_t L = (int64_t)OH * OW;
  const int64_t total = (int64_t)B * C * H * W;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int iw = (int)(p % W);
    const int ih = (int)((p / W) % H);
    const int c = (int)((p / ((int64_t)W * H)) % C);
    const int b = (int)(p / ((int64_t)W * H * C));
    float acc = 0.f;
    for (int kh = 0; kh < K; ++kh) {
      const int num_h = ih + pad - kh;
      if (num_h < 0 || num_h % stride) continue;
      const int oh = num_h / stride;
      if (oh >= OH) continue;
      for (int kw = 0; kw < K; ++kw) {
        const int num_w = iw + pad - kw;
        if (num_w < 0 || num_w % stride) continue;
        const int ow = num_w / stride;
        if (ow >= OW) continue;
        const int row = c * K * K + kh * K + kw;
        acc += bf2f(du[((int64_t)b * ckk_pad + row) * L + oh * OW + ow]);
      }
    }
    dx[p] = f2bf(acc);
  }
}

int ic_grid(int64_t n) {
  return (int)std::min<int64_t>((n + 255) / 256, 4096);
}

}  // namespace

void im2col_bf16(const at::Tensor& x, at::Tensor& out, int64_t K,
                 int64_t stride, int64_t pad) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous() &&
              x.scalar_type() == at::kBFloat16, "im2col: bf16 NCHW");
  const int B = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
            W = (int)x.size(3);
  const int ckk_pad = (int)out.size(1);
  const int OH = (int)out.size(2), OW = (int)out.size(3);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(im2col_bf16_kernel, dim3(ic_grid(out.numel())),
                     dim3(256), 0, stream.stream(),
                     (const bf16_t*)x.data_ptr(), (bf16_t*)out.data_ptr(), B,
                     C, H, W, (int)K, OH, OW, (int)stride, (int)pad,
                     ckk_pad);
  HIP_CHECK_KERNEL();
}

void col2im_bf16(const at::Tensor& du, at::Tensor& dx, int64_t K,
                 int64_t stride, int64_t pad) {
  TORCH_CHECK(du.is_cuda() && du.is_contiguous() && dx.is_contiguous(),
              "col2im: contiguous");
  const int B = (int)dx.size(0), C = (int)dx.size(1), H = (int)dx.size(2),
            W = (int)dx.size(3);
  const int ckk_pad = (int)du.size(1);
  const int OH = (int)du.size(2), OW = (int)du.size(3);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(col2im_bf16_kernel, dim3(ic_grid(dx.numel())),
                     dim3(256), 0, stream.stream(),
                     (const bf16_t*)du.data_ptr(), (bf16_t*)dx.data_ptr(), B,
                     C, H, W, (int)K, OH, OW, (int)stride, (int)pad,
                     ckk_pad);
  HIP_CHECK_KERNEL();
}
