// Weighted-sum-of-logits ensemble mixer (K5) + its backward reductions (K6
// feeds off the same weights on the python side).
//
// The AdaNet ensemble logits F(x) = b + sum_j w_j * h_j(x) (reference
// adanet/ensemble/weighted.py:427-454,545-561) computed as ONE fused kernel
// over the J member logit buffers (frozen members' logits come straight from
// the iteration's HBM cache): out[b,c] = bias[c] + sum_j w_j(*)L_j[b,c],
// where w_j is a scalar (MixtureWeightType.SCALAR) or per-class vector
// (VECTOR). Member buffers are passed as a device pointer table so J is
// runtime-sized without kernel recompiles.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

#define MAXJ 64

__global__ __launch_bounds__(256) void mixer_fwd_kernel(
    const int64_t* __restrict__ ptrs, const float* __restrict__ w,
    const float* __restrict__ bias, bf16_t* __restrict__ out, int J, int B,
    int C, int ldl, int ldo, int vector_mode) {
  const int64_t total = (int64_t)B * C;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int b = (int)(p / C), c = (int)(p % C);
    float acc = bias ? bias[c] : 0.f;
    for (int j = 0; j < J; ++j) {
      const bf16_t* L = (const bf16_t*)ptrs[j];
      const float wj = vector_mode ? w[j * C + c] : w[j];
      acc += wj * bf2f(L[(int64_t)b * ldl + c]);
    }
    out[(int64_t)b * ldo + c] = f2bf(acc);
  }
}

// dw for SCALAR weights: dw[j] = sum_{b,c} dY[b,c] * L_j[b,c].
__global__ __launch_bounds__(256) void mixer_bwd_dw_scalar_kernel(
    const int64_t* __restrict__ ptrs, const bf16_t* __restrict__ dY,
    float* __restrict__ dw, int J, int B, int C, int ldl, int ldy) {
  const int j = blockIdx.y;
  const bf16_t* L = (const bf16_t*)ptrs[j];
  const int64_t total = (int64_t)B * C;
  float acc = 0.f;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int b = (int)(p / C), c = (int)(p % C);
    acc += bf2f(dY[(int64_t)b * ldy + c]) * bf2f(L[(int64_t)b * ldl + c]);
  }
  acc = wave_reduce_sum(acc);
  __shared__ float partial[4];
  if ((threadIdx.x & 63) == 0) partial[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = partial[0] + partial[1] + partial[2] + partial[3];
    atomicAdd(&dw[j], s);
  }
}

// dw for VECTOR weights: dw[j][c] = sum_b dY[b,c] * L_j[b,c].
// One thread per (j, c); loops rows (column reads ride the L2/L3: the
// mixer tensors are tiny next to the 256 MiB Infinity Cache).
__global__ __launch_bounds__(256) void mixer_bwd_dw_vector_kernel(
    const int64_t* __restrict__ ptrs, const bf16_t* __restrict__ dY,
    float* __restrict__ dw, int J, int B, int C, int ldl, int ldy) {
  const int j = blockIdx.y;
  const bf16_t* L = (const bf16_t*)ptrs[j];
  for (int c = blockIdx.x * blockDim.x + threadIdx.x; c < C;
       c += gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int b = 0; b < B; ++b)
      acc += bf2f(dY[(int64_t)b * ldy + c]) * bf2f(L[(int64_t)b * ldl + c]);
    dw[j * C + c] = acc;
  }
}

// dL_j = w_j * dY (only trainable members need it; frozen logits take no grad).
__global__ __launch_bounds__(256) void mixer_bwd_dlogits_kernel(
    const bf16_t* __restrict__ dY, const float* __restrict__ w,
    bf16_t* __restrict__ dL, int B, int C, int ldy, int ldl, int j,
    int vector_mode) {
  const int64_t total = (int64_t)B * C;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int b = (int)(p / C), c = (int)(p % C);
    const float wj = vector_mode ? w[j * C + c] : w[j];
    dL[(int64_t)b * ldl + c] = f2bf(wj * bf2f(dY[(int64_t)b * ldy + c]));
  }
}

static int grid_for(int64_t total) {
  return (int)std::min<int64_t>((total + 255) / 256, 2048);
}

void mixer_fwd(const at::Tensor& ptrs, const at::Tensor& weights,
               const c10::optional<at::Tensor>& bias, at::Tensor& out,
               int64_t B, int64_t C, int64_t ldl, int64_t vector_mode) {
  const int J = (int)ptrs.numel();
  TORCH_CHECK(J <= MAXJ, "mixer: too many members");
  auto stream = at::cuda::getCurrentCUDAStream();
  const float* bias_ptr =
      (bias.has_value() && bias->defined()) ? bias->data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(mixer_fwd_kernel, dim3(grid_for(B * C)), dim3(256), 0,
                     stream.stream(), ptrs.data_ptr<int64_t>(),
                     weights.data_ptr<float>(), bias_ptr,
                     (bf16_t*)out.data_ptr(), J, (int)B, (int)C, (int)ldl,
                     (int)out.stride(0), (int)vector_mode);
  HIP_CHECK_KERNEL();
}

void mixer_bwd_dw(const at::Tensor& ptrs, const at::Tensor& dY, at::Tensor& dw,
                  int64_t B, int64_t C, int64_t ldl, int64_t vector_mode) {
  const int J = (int)ptrs.numel();
  auto stream = at::cuda::getCurrentCUDAStream();
  if (vector_mode) {
    dim3 grid((unsigned)((C + 255) / 256), (unsigned)J);
    hipLaunchKernelGGL(mixer_bwd_dw_vector_kernel, grid, dim3(256), 0,
                       stream.stream(), ptrs.data_ptr<int64_t>(),
                       (const bf16_t*)dY.data_ptr(), dw.data_ptr<float>(), J,
                       (int)B, (int)C, (int)ldl, (int)dY.stride(0));
  } else {
    dim3 grid((unsigned)std::min<int64_t>((B * C + 2047) / 2048, 256),
              (unsigned)J);
    hipLaunchKernelGGL(mixer_bwd_dw_scalar_kernel, grid, dim3(256), 0,
                       stream.stream(), ptrs.data_ptr<int64_t>(),
                       (const bf16_t*)dY.data_ptr(), dw.data_ptr<float>(), J,
                       (int)B, (int)C, (int)ldl, (int)dY.stride(0));
  }
  HIP_CHECK_KERNEL();
}

void mixer_bwd_dlogits(const at::Tensor& dY, const at::Tensor& weights,
                       at::Tensor& dL, int64_t j, int64_t vector_mode) {
  const int B = (int)dY.size(0), C = (int)dY.size(1);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(mixer_bwd_dlogits_kernel, dim3(grid_for((int64_t)B * C)),
                     dim3(256), 0, stream.stream(),
                     (const bf16_t*)dY.data_ptr(), weights.data_ptr<float>(),
                     (bf16_t*)dL.data_ptr(), B, C, (int)dY.stride(0),
                     (int)dL.stride(0), (int)j, (int)vector_mode);
  HIP_CHECK_KERNEL();
}
