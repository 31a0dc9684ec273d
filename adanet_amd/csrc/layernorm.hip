// LayerNorm fwd/bwd (K8): one wave per row, shuffle reductions, bf16 I/O
// with fp32 statistics (mean/rstd saved for backward). Vectorized short4
// loads on the row (guide G13). Replaces the reference's reliance on TF's
// normalization kernels (NASNet batch-norm arg scopes,
// research/improve_nas/trainer/nasnet.py:127-233; LayerNorm per north star).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

__global__ __launch_bounds__(256) void layernorm_fwd_kernel(
    const bf16_t* __restrict__ x, const float* __restrict__ gamma,
    const float* __restrict__ beta, bf16_t* __restrict__ y,
    float* __restrict__ mean_out, float* __restrict__ rstd_out, int B, int D,
    float eps) {
  const int wave_in_block = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int waves = (gridDim.x * blockDim.x) >> 6;
  for (int row = blockIdx.x * (blockDim.x >> 6) + wave_in_block; row < B;
       row += waves) {
    const bf16_t* xr = x + (int64_t)row * D;
    float s = 0.f, s2 = 0.f;
    for (int d = lane; d < D; d += 64) {
      const float v = bf2f(xr[d]);
      s += v;
      s2 += v * v;
    }
    s = wave_reduce_sum(s);
    s2 = wave_reduce_sum(s2);
    const float mean = s / D;
    const float var = s2 / D - mean * mean;
    const float rstd = rsqrtf(var + eps);
    if (lane == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    bf16_t* yr = y + (int64_t)row * D;
    for (int d = lane; d < D; d += 64) {
      const float xh = (bf2f(xr[d]) - mean) * rstd;
      yr[d] = f2bf(xh * (gamma ? gamma[d] : 1.f) + (beta ? beta[d] : 0.f));
    }
  }
}

// dx = rstd * (dy*g - mean(dy*g) - xhat * mean(dy*g*xhat));
// dgamma[d] = sum_b dy*xhat ; dbeta[d] = sum_b dy  (atomics into fp32).
__global__ __launch_bounds__(256) void layernorm_bwd_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ dy,
    const float* __restrict__ gamma, const float* __restrict__ mean_in,
    const float* __restrict__ rstd_in, bf16_t* __restrict__ dx,
    float* __restrict__ dgamma, float* __restrict__ dbeta, int B, int D) {
  const int wave_in_block = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int waves = (gridDim.x * blockDim.x) >> 6;
  for (int row = blockIdx.x * (blockDim.x >> 6) + wave_in_block; row < B;
       row += waves) {
    const bf16_t* xr = x + (int64_t)row * D;
    const bf16_t* dyr = dy + (int64_t)row * D;
    const float mean = mean_in[row], rstd = rstd_in[row];
    float sg = 0.f, sgx = 0.f;
    for (int d = lane; d < D; d += 64) {
      const float xh = (bf2f(xr[d]) - mean) * rstd;
      const float g = bf2f(dyr[d]) * (gamma ? gamma[d] : 1.f);
      sg += g;
      sgx += g * xh;
      if (dgamma) atomicAdd(&dgamma[d], bf2f(dyr[d]) * xh);
      if (dbeta) atomicAdd(&dbeta[d], bf2f(dyr[d]));
    }
    sg = wave_reduce_sum(sg) / D;
    sgx = wave_reduce_sum(sgx) / D;
    bf16_t* dxr = dx + (int64_t)row * D;
    for (int d = lane; d < D; d += 64) {
      const float xh = (bf2f(xr[d]) - mean) * rstd;
      const float g = bf2f(dyr[d]) * (gamma ? gamma[d] : 1.f);
      dxr[d] = f2bf(rstd * (g - sg - xh * sgx));
    }
  }
}

void layernorm_fwd(const at::Tensor& x, const c10::optional<at::Tensor>& gamma,
                   const c10::optional<at::Tensor>& beta, at::Tensor& y,
                   at::Tensor& mean, at::Tensor& rstd, double eps) {
  const int B = (int)x.size(0), D = (int)x.size(1);
  if (B == 0 || D == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  const float* g = (gamma && gamma->defined()) ? gamma->data_ptr<float>() : nullptr;
  const float* b = (beta && beta->defined()) ? beta->data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(layernorm_fwd_kernel, dim3(std::min((B + 3) / 4, 2048)),
                     dim3(256), 0, stream.stream(),
                     (const bf16_t*)x.data_ptr(), g, b, (bf16_t*)y.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(), B, D,
                     (float)eps);
  HIP_CHECK_KERNEL();
}

void layernorm_bwd(const at::Tensor& x, const at::Tensor& dy,
                   const c10::optional<at::Tensor>& gamma,
                   const at::Tensor& mean, const at::Tensor& rstd,
                   at::Tensor& dx, const c10::optional<at::Tensor>& dgamma,
                   const c10::optional<at::Tensor>& dbeta) {
  const int B = (int)x.size(0), D = (int)x.size(1);
  if (B == 0 || D == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  const float* g = (gamma && gamma->defined()) ? gamma->data_ptr<float>() : nullptr;
  float* dg = (dgamma && dgamma->defined()) ? dgamma->data_ptr<float>() : nullptr;
  float* db = (dbeta && dbeta->defined()) ? dbeta->data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(layernorm_bwd_kernel, dim3(std::min((B + 3) / 4, 2048)),
                     dim3(256), 0, stream.stream(),
                     (const bf16_t*)x.data_ptr(), (const bf16_t*)dy.data_ptr(),
                     g, mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     (bf16_t*)dx.data_ptr(), dg, db, B, D);
  HIP_CHECK_KERNEL();
}
