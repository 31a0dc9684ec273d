// BatchNorm2d fwd/bwd (K8 remainder): per-channel statistics over N*H*W,
// bf16 NCHW activations with fp32 stats/affine — replaces the round-1
// torch nn.BatchNorm2d fallback in the NASNet cells (reference arg scopes
// research/improve_nas/trainer/nasnet.py:127-233).
//
// Structure: one 256-thread block per channel for the reductions
// (wave shuffle + LDS cross-wave, vectorized short8 row loads when HW%8==0)
// and a grid-stride elementwise kernel for normalize / dx. dgamma == sum
// dy*xhat and dbeta == sum dy fall out of the backward reduction for free.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

struct bn8 {
  bf16_t v[8];
};

__device__ __forceinline__ float block_reduce(float v, float* scratch,
                                              int tid) {
  v = wave_reduce_sum(v);
  const int wid = tid >> 6, lane = tid & 63;
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float out = (tid < 4) ? scratch[tid] : 0.f;
  if (tid == 0) {
    out = scratch[0] + scratch[1] + scratch[2] + scratch[3];
    scratch[0] = out;
  }
  __syncthreads();
  out = scratch[0];
  __syncthreads();
  return out;
}

// Chunked partial sums: grid (C, nchunks) so small-C shapes still fill
// the chip (a C-block grid was 30% of the improve_nas step on 32 CUs,
// profiles/nasprof3_summary.txt). Each block writes its OWN workspace
// slot wsp[chunk][2][C]; the finalize kernel sums chunks in fixed order —
// deterministic, and the workspace needs no zero-fill (the at::zeros +
// .zero_() launches were 2.9% of the step, profiles/nasprof6_summary).
__global__ __launch_bounds__(256) void bn_stats_partial_kernel(
    const bf16_t* __restrict__ x, float* __restrict__ wsp, int N, int C,
    int64_t HW, int nchunks) {
  __shared__ float scratch[4];
  const int c = blockIdx.x;
  const int n0 = (N * blockIdx.y) / nchunks;
  const int n1 = (N * (blockIdx.y + 1)) / nchunks;
  const int tid = threadIdx.x;
  float s = 0.f, s2 = 0.f;
  for (int n = n0; n < n1; ++n) {
    const bf16_t* base = x + ((int64_t)n * C + c) * HW;
    if ((HW & 7) == 0) {
      for (int64_t i = (int64_t)tid * 8; i < HW; i += 256 * 8) {
        const bn8 v = *(const bn8*)(base + i);
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          const float f = bf2f(v.v[k]);
          s += f;
          s2 += f * f;
        }
      }
    } else {
      for (int64_t i = tid; i < HW; i += 256) {
        const float f = bf2f(base[i]);
        s += f;
        s2 += f * f;
      }
    }
  }
  s = block_reduce(s, scratch, tid);
  s2 = block_reduce(s2, scratch, tid);
  if (tid == 0) {
    float* slot = wsp + (int64_t)blockIdx.y * 2 * C;
    slot[c] = s;
    slot[C + c] = s2;
  }
}

// One wave per channel, lanes over chunks (fixed-tree reduce — the
// thread-per-channel loop serialized nchunks loads on a near-empty
// chip: 7 us/call, 1.2% of the NASNet step, profiles/nasprof8).
__global__ __launch_bounds__(256) void bn_stats_finalize_kernel(
    const float* __restrict__ wsp, float* __restrict__ mean_out,
    float* __restrict__ rstd_out, float* __restrict__ running_mean,
    float* __restrict__ running_var, int N, int C, int64_t HW, float eps,
    float momentum, int nchunks) {
  const int c = blockIdx.x * 4 + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (c >= C) return;
  float s = 0.f, s2 = 0.f;
  for (int r = lane; r < nchunks; r += 64) {
    const float* slot = wsp + (int64_t)r * 2 * C;
    s += slot[c];
    s2 += slot[C + c];
  }
  s = wave_reduce_sum(s);
  s2 = wave_reduce_sum(s2);
  if (lane != 0) return;
  const double M = (double)N * (double)HW;
  const float mean = (float)(s / M);
  float var = (float)(s2 / M) - mean * mean;
  var = var > 0.f ? var : 0.f;
  mean_out[c] = mean;
  rstd_out[c] = rsqrtf(var + eps);
  if (running_mean) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
    const float unbiased = M > 1.0 ? var * (float)(M / (M - 1.0)) : var;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// Per-channel mean/rstd (+ running-stat update). Block c handles channel c.
__global__ __launch_bounds__(256) void bn_stats_kernel(
    const bf16_t* __restrict__ x, float* __restrict__ mean_out,
    float* __restrict__ rstd_out, float* __restrict__ running_mean,
    float* __restrict__ running_var, int N, int C, int64_t HW, float eps,
    float momentum) {
  __shared__ float scratch[4];
  const int c = blockIdx.x;
  const int tid = threadIdx.x;
  float s = 0.f, s2 = 0.f;
  for (int n = 0; n < N; ++n) {
    const bf16_t* base = x + ((int64_t)n * C + c) * HW;
    if ((HW & 7) == 0) {
      for (int64_t i = (int64_t)tid * 8; i < HW; i += 256 * 8) {
        const bn8 v = *(const bn8*)(base + i);
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          const float f = bf2f(v.v[k]);
          s += f;
          s2 += f * f;
        }
      }
    } else {
      for (int64_t i = tid; i < HW; i += 256) {
        const float f = bf2f(base[i]);
        s += f;
        s2 += f * f;
      }
    }
  }
  s = block_reduce(s, scratch, tid);
  s2 = block_reduce(s2, scratch, tid);
  if (tid == 0) {
    const double M = (double)N * (double)HW;
    const float mean = (float)(s / M);
    float var = (float)(s2 / M) - mean * mean;
    var = var > 0.f ? var : 0.f;
    mean_out[c] = mean;
    rstd_out[c] = rsqrtf(var + eps);
    if (running_mean) {
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
      const float unbiased = M > 1.0 ? var * (float)(M / (M - 1.0)) : var;
      running_var[c] = (1.f - momentum) * running_var[c] +
                       momentum * unbiased;
    }
  }
}

// y = (x - mean[c]) * rstd[c] * gamma[c] + beta[c]
__global__ __launch_bounds__(256) void bn_norm_kernel(
    const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta, int C,
    int64_t HW, int64_t total) {
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int c = (int)((p / HW) % C);
    const float g = gamma ? gamma[c] : 1.f;
    const float b = beta ? beta[c] : 0.f;
    y[p] = f2bf((bf2f(x[p]) - mean[c]) * rstd[c] * g + b);
  }
}

// Chunked/vectorized variant of the backward reduce: grid (C, nchunks),
// per-chunk workspace slots wsp[chunk][2][C] + fixed-order finalize
// (deterministic; no host zero-fill).
__global__ __launch_bounds__(256) void bn_bwd_reduce_chunked_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ dy,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    float* __restrict__ wsp, int N, int C, int64_t HW, int nchunks) {
  __shared__ float scratch[4];
  const int c = blockIdx.x;
  const int n0 = (N * blockIdx.y) / nchunks;
  const int n1 = (N * (blockIdx.y + 1)) / nchunks;
  const int tid = threadIdx.x;
  const float m = mean[c], r = rstd[c];
  float a = 0.f, b = 0.f;
  for (int n = n0; n < n1; ++n) {
    const int64_t off = ((int64_t)n * C + c) * HW;
    if ((HW & 7) == 0) {
      for (int64_t i = (int64_t)tid * 8; i < HW; i += 256 * 8) {
        const bn8 dv = *(const bn8*)(dy + off + i);
        const bn8 xv = *(const bn8*)(x + off + i);
#pragma unroll
        for (int k = 0; k < 8; ++k) {
          const float d = bf2f(dv.v[k]);
          a += d;
          b += d * (bf2f(xv.v[k]) - m) * r;
        }
      }
    } else {
      for (int64_t i = tid; i < HW; i += 256) {
        const float d = bf2f(dy[off + i]);
        a += d;
        b += d * (bf2f(x[off + i]) - m) * r;
      }
    }
  }
  a = block_reduce(a, scratch, tid);
  b = block_reduce(b, scratch, tid);
  if (tid == 0) {
    float* slot = wsp + (int64_t)blockIdx.y * 2 * C;
    slot[c] = a;
    slot[C + c] = b;
  }
}

// sdy/sdyx[c] = sum over chunks: wave per channel, lanes over chunks.
__global__ __launch_bounds__(256) void bn_bwd_finalize_kernel(
    const float* __restrict__ wsp, float* __restrict__ sdy,
    float* __restrict__ sdyx, int C, int nchunks) {
  const int c = blockIdx.x * 4 + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (c >= C) return;
  float a = 0.f, b = 0.f;
  for (int r = lane; r < nchunks; r += 64) {
    const float* slot = wsp + (int64_t)r * 2 * C;
    a += slot[c];
    b += slot[C + c];
  }
  a = wave_reduce_sum(a);
  b = wave_reduce_sum(b);
  if (lane == 0) {
    sdy[c] = a;
    sdyx[c] = b;
  }
}

// Per-channel backward sums: sdy = sum dy (== dbeta), sdyx = sum dy*xhat
// (== dgamma).
__global__ __launch_bounds__(256) void bn_bwd_reduce_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ dy,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    float* __restrict__ sdy, float* __restrict__ sdyx, int N, int C,
    int64_t HW) {
  __shared__ float scratch[4];
  const int c = blockIdx.x;
  const int tid = threadIdx.x;
  const float m = mean[c], r = rstd[c];
  float a = 0.f, b = 0.f;
  for (int n = 0; n < N; ++n) {
    const int64_t off = ((int64_t)n * C + c) * HW;
    for (int64_t i = tid; i < HW; i += 256) {
      const float d = bf2f(dy[off + i]);
      const float xh = (bf2f(x[off + i]) - m) * r;
      a += d;
      b += d * xh;
    }
  }
  a = block_reduce(a, scratch, tid);
  b = block_reduce(b, scratch, tid);
  if (tid == 0) {
    sdy[c] = a;
    sdyx[c] = b;
  }
}

// dx = gamma*rstd * (dy - sdy/M - xhat * sdyx/M)
__global__ __launch_bounds__(256) void bn_bwd_dx_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ dy,
    bf16_t* __restrict__ dx, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ gamma,
    const float* __restrict__ sdy, const float* __restrict__ sdyx, int C,
    int64_t HW, int64_t total, float invM) {
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int c = (int)((p / HW) % C);
    const float g = gamma ? gamma[c] : 1.f;
    const float xh = (bf2f(x[p]) - mean[c]) * rstd[c];
    dx[p] = f2bf(g * rstd[c] *
                 (bf2f(dy[p]) - sdy[c] * invM - xh * sdyx[c] * invM));
  }
}

int grid_for(int64_t total) {
  return (int)std::min<int64_t>((total + 255) / 256, 2048);
}

}  // namespace

void batchnorm_stats(const at::Tensor& x, at::Tensor& mean, at::Tensor& rstd,
                     const c10::optional<at::Tensor>& running_mean,
                     const c10::optional<at::Tensor>& running_var,
                     double eps, double momentum) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous() &&
              x.scalar_type() == at::kBFloat16, "bn_stats: bf16 NCHW");
  const int N = (int)x.size(0), C = (int)x.size(1);
  const int64_t HW = x.size(2) * x.size(3);
  auto stream = at::cuda::getCurrentCUDAStream();
  float* rm = nullptr;
  float* rv = nullptr;
  if (running_mean.has_value() && running_mean->defined()) {
    rm = running_mean->data_ptr<float>();
    rv = running_var->data_ptr<float>();
  }
  const int nchunks = (int)std::max<int64_t>(
      1, std::min<int64_t>(2048 / C, std::min<int64_t>(N, 64)));
  if (nchunks > 1) {
    auto wsp = at::empty({nchunks, 2, C}, x.options().dtype(at::kFloat));
    hipLaunchKernelGGL(bn_stats_partial_kernel, dim3(C, nchunks), dim3(256),
                       0, stream.stream(), (const bf16_t*)x.data_ptr(),
                       wsp.data_ptr<float>(), N, C, HW, nchunks);
    hipLaunchKernelGGL(bn_stats_finalize_kernel, dim3((C + 3) / 4),
                       dim3(256), 0, stream.stream(), wsp.data_ptr<float>(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), rm,
                       rv, N, C, HW, (float)eps, (float)momentum, nchunks);
    HIP_CHECK_KERNEL();
    return;
  }
  hipLaunchKernelGGL(bn_stats_kernel, dim3(C), dim3(256), 0, stream.stream(),
                     (const bf16_t*)x.data_ptr(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), rm, rv, N, C, HW, (float)eps,
                     (float)momentum);
  HIP_CHECK_KERNEL();
}

void batchnorm_norm(const at::Tensor& x, at::Tensor& y,
                    const at::Tensor& mean, const at::Tensor& rstd,
                    const c10::optional<at::Tensor>& gamma,
                    const c10::optional<at::Tensor>& beta) {
  const int C = (int)x.size(1);
  const int64_t HW = x.size(2) * x.size(3);
  const int64_t total = x.numel();
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(
      bn_norm_kernel, dim3(grid_for(total)), dim3(256), 0, stream.stream(),
      (const bf16_t*)x.data_ptr(), (bf16_t*)y.data_ptr(),
      mean.data_ptr<float>(), rstd.data_ptr<float>(),
      gamma.has_value() && gamma->defined() ? gamma->data_ptr<float>()
                                            : nullptr,
      beta.has_value() && beta->defined() ? beta->data_ptr<float>() : nullptr,
      C, HW, total);
  HIP_CHECK_KERNEL();
}

void batchnorm_bwd(const at::Tensor& x, const at::Tensor& dy, at::Tensor& dx,
                   const at::Tensor& mean, const at::Tensor& rstd,
                   const c10::optional<at::Tensor>& gamma, at::Tensor& sdy,
                   at::Tensor& sdyx) {
  const int N = (int)x.size(0), C = (int)x.size(1);
  const int64_t HW = x.size(2) * x.size(3);
  const int64_t total = x.numel();
  auto stream = at::cuda::getCurrentCUDAStream();
  const int nchunks = (int)std::max<int64_t>(
      1, std::min<int64_t>(2048 / C, std::min<int64_t>(N, 64)));
  if (nchunks > 1) {
    auto wsp = at::empty({nchunks, 2, C}, sdy.options());
    hipLaunchKernelGGL(bn_bwd_reduce_chunked_kernel, dim3(C, nchunks),
                       dim3(256), 0, stream.stream(),
                       (const bf16_t*)x.data_ptr(),
                       (const bf16_t*)dy.data_ptr(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), wsp.data_ptr<float>(), N, C,
                       HW, nchunks);
    hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3((C + 3) / 4),
                       dim3(256), 0, stream.stream(), wsp.data_ptr<float>(),
                       sdy.data_ptr<float>(), sdyx.data_ptr<float>(), C,
                       nchunks);
  } else {
    hipLaunchKernelGGL(bn_bwd_reduce_kernel, dim3(C), dim3(256), 0,
                       stream.stream(), (const bf16_t*)x.data_ptr(),
                       (const bf16_t*)dy.data_ptr(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), sdy.data_ptr<float>(),
                       sdyx.data_ptr<float>(), N, C, HW);
  }
  HIP_CHECK_KERNEL();
  const float invM = 1.f / (float)((double)N * (double)HW);
  hipLaunchKernelGGL(
      bn_bwd_dx_kernel, dim3(grid_for(total)), dim3(256), 0, stream.stream(),
      (const bf16_t*)x.data_ptr(), (const bf16_t*)dy.data_ptr(),
      (bf16_t*)dx.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
      gamma.has_value() && gamma->defined() ? gamma->data_ptr<float>()
                                            : nullptr,
      sdy.data_ptr<float>(), sdyx.data_ptr<float>(), C, HW, total, invM);
  HIP_CHECK_KERNEL();
}
