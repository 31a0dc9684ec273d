// Fused optimizers (K4): single-pass SGD(+momentum) and Adam over FLAT
// parameter arenas — the engine flattens each candidate's parameters into
// one contiguous bf16 working buffer + fp32 master copy (+ fp32 optimizer
// state), so the whole update is ONE kernel launch per candidate and the
// distributed gradient all-reduce is ONE RCCL bucket (reference dependency:
// optimizer.minimize() per candidate, adanet/examples/simple_dnn.py:110,
// research/improve_nas/trainer/optimizer.py:83-135).
//
// Update math (fp32 master):
//   g = grad / grad_scale (+ weight_decay * p)
//   momentum: m = mu*m + (1-dampening)*g ; p -= lr * (nesterov ? g + mu*m : m)
//   adam:     m = b1*m+(1-b1)g ; v = b2*v+(1-b2)g^2 ;
//             p -= lr * mhat / (sqrt(vhat)+eps)
// then the bf16 working copy is refreshed from the master in the same pass.
// Vectorized 4-wide; memory-bound by design (guide G13).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

__global__ __launch_bounds__(256) void fused_sgd_kernel(
    float* __restrict__ master, bf16_t* __restrict__ param,
    const bf16_t* __restrict__ grad, float* __restrict__ mom, int64_t n,
    float lr, float mu, float dampening, float weight_decay, int nesterov,
    float inv_scale) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float p = master[i];
    float g = bf2f(grad[i]) * inv_scale + weight_decay * p;
    if (mom != nullptr) {
      float m = mom[i] * mu + (1.f - dampening) * g;
      mom[i] = m;
      g = nesterov ? g + mu * m : m;
    }
    p -= lr * g;
    master[i] = p;
    param[i] = f2bf(p);
  }
}

__global__ __launch_bounds__(256) void fused_adam_kernel(
    float* __restrict__ master, bf16_t* __restrict__ param,
    const bf16_t* __restrict__ grad, float* __restrict__ m_buf,
    float* __restrict__ v_buf, int64_t n, float lr, float beta1, float beta2,
    float eps, float weight_decay, float bc1, float bc2, float inv_scale) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float p = master[i];
    float g = bf2f(grad[i]) * inv_scale + weight_decay * p;
    float m = beta1 * m_buf[i] + (1.f - beta1) * g;
    float v = beta2 * v_buf[i] + (1.f - beta2) * g * g;
    m_buf[i] = m;
    v_buf[i] = v;
    const float mhat = m / bc1;
    const float vhat = v / bc2;
    p -= lr * mhat / (__builtin_sqrtf(vhat) + eps);
    master[i] = p;
    param[i] = f2bf(p);
  }
}

// fp32-native variants (mixture weights / biases): param IS the master.
__global__ __launch_bounds__(256) void fused_sgd_fp32_kernel(
    float* __restrict__ param, const float* __restrict__ grad,
    float* __restrict__ mom, int64_t n, float lr, float mu, float dampening,
    float weight_decay, int nesterov, float inv_scale) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float p = param[i];
    float g = grad[i] * inv_scale + weight_decay * p;
    if (mom != nullptr) {
      float m = mom[i] * mu + (1.f - dampening) * g;
      mom[i] = m;
      g = nesterov ? g + mu * m : m;
    }
    param[i] = p - lr * g;
  }
}

__global__ __launch_bounds__(256) void fused_adam_fp32_kernel(
    float* __restrict__ param, const float* __restrict__ grad,
    float* __restrict__ m_buf, float* __restrict__ v_buf, int64_t n, float lr,
    float beta1, float beta2, float eps, float weight_decay, float bc1,
    float bc2, float inv_scale) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float p = param[i];
    float g = grad[i] * inv_scale + weight_decay * p;
    float m = beta1 * m_buf[i] + (1.f - beta1) * g;
    float v = beta2 * v_buf[i] + (1.f - beta2) * g * g;
    m_buf[i] = m;
    v_buf[i] = v;
    param[i] = p - lr * (m / bc1) / (__builtin_sqrtf(v / bc2) + eps);
  }
}

static int opt_grid(int64_t n) {
  return (int)std::min<int64_t>((n + 255) / 256, 2048);
}

void fused_sgd_fp32(at::Tensor& param, const at::Tensor& grad,
                    const c10::optional<at::Tensor>& momentum_buf, double lr,
                    double momentum, double dampening, double weight_decay,
                    bool nesterov, double grad_scale) {
  const int64_t n = param.numel();
  auto stream = at::cuda::getCurrentCUDAStream();
  float* mom = (momentum_buf.has_value() && momentum_buf->defined())
                   ? momentum_buf->data_ptr<float>()
                   : nullptr;
  hipLaunchKernelGGL(fused_sgd_fp32_kernel, dim3(opt_grid(n)), dim3(256), 0,
                     stream.stream(), param.data_ptr<float>(),
                     grad.data_ptr<float>(), mom, n, (float)lr,
                     (float)momentum, (float)dampening, (float)weight_decay,
                     nesterov ? 1 : 0, (float)(1.0 / grad_scale));
  HIP_CHECK_KERNEL();
}

void fused_adam_fp32(at::Tensor& param, const at::Tensor& grad,
                     at::Tensor& m_buf, at::Tensor& v_buf, double lr,
                     double beta1, double beta2, double eps,
                     double weight_decay, int64_t step, double grad_scale) {
  const int64_t n = param.numel();
  auto stream = at::cuda::getCurrentCUDAStream();
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);
  hipLaunchKernelGGL(fused_adam_fp32_kernel, dim3(opt_grid(n)), dim3(256), 0,
                     stream.stream(), param.data_ptr<float>(),
                     grad.data_ptr<float>(), m_buf.data_ptr<float>(),
                     v_buf.data_ptr<float>(), n, (float)lr, (float)beta1,
                     (float)beta2, (float)eps, (float)weight_decay, bc1, bc2,
                     (float)(1.0 / grad_scale));
  HIP_CHECK_KERNEL();
}

void fused_sgd(at::Tensor& master, at::Tensor& param, const at::Tensor& grad,
               const c10::optional<at::Tensor>& momentum_buf, double lr,
               double momentum, double dampening, double weight_decay,
               bool nesterov, double grad_scale) {
  const int64_t n = master.numel();
  if (n == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  float* mom = (momentum_buf.has_value() && momentum_buf->defined())
                   ? momentum_buf->data_ptr<float>()
                   : nullptr;
  hipLaunchKernelGGL(fused_sgd_kernel, dim3(opt_grid(n)), dim3(256), 0,
                     stream.stream(), master.data_ptr<float>(),
                     (bf16_t*)param.data_ptr(), (const bf16_t*)grad.data_ptr(),
                     mom, n, (float)lr, (float)momentum, (float)dampening,
                     (float)weight_decay, nesterov ? 1 : 0,
                     (float)(1.0 / grad_scale));
  HIP_CHECK_KERNEL();
}

void fused_adam(at::Tensor& master, at::Tensor& param, const at::Tensor& grad,
                at::Tensor& m_buf, at::Tensor& v_buf, double lr, double beta1,
                double beta2, double eps, double weight_decay, int64_t step,
                double grad_scale) {
  const int64_t n = master.numel();
  auto stream = at::cuda::getCurrentCUDAStream();
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);
  hipLaunchKernelGGL(fused_adam_kernel, dim3(opt_grid(n)), dim3(256), 0,
                     stream.stream(), master.data_ptr<float>(),
                     (bf16_t*)param.data_ptr(), (const bf16_t*)grad.data_ptr(),
                     m_buf.data_ptr<float>(), v_buf.data_ptr<float>(), n,
                     (float)lr, (float)beta1, (float)beta2, (float)eps,
                     (float)weight_decay, bc1, bc2, (float)(1.0 / grad_scale));
  HIP_CHECK_KERNEL();
}
