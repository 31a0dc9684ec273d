// Weighted-sum-of-logits ensemble mixer (K5) + its backward reductions.
//
// The AdaNet ensemble logits F(x) = b + sum_j w_j * h_j(x) (reference
// adanet/ensemble/weighted.py:427-454,545-561) computed as ONE fused kernel
// over the J member logit buffers: out[b,c] = bias[c] + sum_j w_j(*)L[j,b,c],
// where w_j is a scalar (MixtureWeightType.SCALAR) or per-class vector
// (VECTOR). The production path reads the J member buffers IN PLACE via
// kernel-arg pointer structs (mixer_*_direct below — no per-step stack
// copy; hipGraph capture bakes the stable static-buffer addresses). The
// older stacked-[J,B,C] entry points are kept for the probe/tests.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

__global__ __launch_bounds__(256) void mixer_fwd_kernel(
    const bf16_t* __restrict__ stack, const float* __restrict__ w,
    const float* __restrict__ bias, bf16_t* __restrict__ out, int J, int B,
    int C, int64_t stride_j, bf16_t* __restrict__ outdst_unused,
    int ldo, int vector_mode) {
  const int64_t total = (int64_t)B * C;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int b = (int)(p / C), c = (int)(p % C);
    float acc = bias ? bias[c] : 0.f;
    const int64_t off = (int64_t)b * C + c;
    for (int j = 0; j < J; ++j) {
      const float wj = vector_mode ? w[j * C + c] : w[j];
      acc += wj * bf2f(stack[j * stride_j + off]);
    }
    out[(int64_t)b * ldo + c] = f2bf(acc);
  }
}

// dw for SCALAR weights: dw[j] = sum_{b,c} dY[b,c] * L[j,b,c].
__global__ __launch_bounds__(256) void mixer_bwd_dw_scalar_kernel(
    const bf16_t* __restrict__ stack, const bf16_t* __restrict__ dY,
    float* __restrict__ wsp, int J, int B, int C, int64_t stride_j,
    int ldy) {
  const int j = blockIdx.y;
  const bf16_t* L = stack + j * stride_j;
  const int64_t total = (int64_t)B * C;
  float acc = 0.f;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int b = (int)(p / C), c = (int)(p % C);
    acc += bf2f(dY[(int64_t)b * ldy + c]) * bf2f(L[(int64_t)b * C + c]);
  }
  acc = wave_reduce_sum(acc);
  __shared__ float partial[4];
  if ((threadIdx.x & 63) == 0) partial[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    wsp[(int64_t)j * gridDim.x + blockIdx.x] =
        partial[0] + partial[1] + partial[2] + partial[3];
  }
}

// dw[j] += sum_b wsp[j][b] in fixed block order (dw holds the running
// arena gradient — accumulate, never overwrite).
__global__ void mixer_dw_reduce_kernel(const float* __restrict__ wsp,
                                       float* __restrict__ dw, int n,
                                       int nblocks) {
  const int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= n) return;
  float s = 0.f;
  for (int b = 0; b < nblocks; ++b) s += wsp[(int64_t)j * nblocks + b];
  dw[j] += s;
}

// dw for VECTOR weights: dw[j][c] = sum_b dY[b,c] * L[j,b,c].
__global__ __launch_bounds__(256) void mixer_bwd_dw_vector_kernel(
    const bf16_t* __restrict__ stack, const bf16_t* __restrict__ dY,
    float* __restrict__ dw, int J, int B, int C, int64_t stride_j, int ldy) {
  const int j = blockIdx.y;
  const bf16_t* L = stack + j * stride_j;
  for (int c = blockIdx.x * blockDim.x + threadIdx.x; c < C;
       c += gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int b = 0; b < B; ++b)
      acc += bf2f(dY[(int64_t)b * ldy + c]) * bf2f(L[(int64_t)b * C + c]);
    dw[j * C + c] = acc;
  }
}

// dL_j = w_j * dY (only trainable members need it).
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

// Direct-pointer variants: members are read in place (frozen static
// buffers + the live candidate's logits) — no per-step [J,B,C] stack copy.
// Pointers ride in the kernel-arg struct, so hipGraph capture bakes the
// STATIC buffer addresses (stable across replays), unlike the host-built
// pointer-table-in-memory approach that captured stale H2D copies.
// J > 8 chunks through repeated launches (accum=1 adds into out).
struct MixerPtrs {
  const bf16_t* p[8];
  int ld[8];
};

__global__ __launch_bounds__(256) void mixer_fwd_ptrs_kernel(
    MixerPtrs Ls, const float* __restrict__ w,
    const float* __restrict__ bias, bf16_t* __restrict__ out, int J, int B,
    int C, int ldo, int vector_mode, int accum) {
  const int64_t total = (int64_t)B * C;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int b = (int)(p / C), c = (int)(p % C);
    bf16_t* op = out + (int64_t)b * ldo + c;
    float acc = accum ? bf2f(*op) : (bias ? bias[c] : 0.f);
    for (int j = 0; j < J; ++j) {
      const float wj = vector_mode ? w[j * C + c] : w[j];
      acc += wj * bf2f(Ls.p[j][(int64_t)b * Ls.ld[j] + c]);
    }
    *op = f2bf(acc);
  }
}

// Deterministic: each block writes its partial into wsp[j][blockIdx.x];
// mixer_dw_reduce_kernel sums in fixed block order (fp32 atomics had
// run-dependent ordering -> last-ulp dw wobble across identical runs).
__global__ __launch_bounds__(256) void mixer_bwd_dw_scalar_ptrs_kernel(
    MixerPtrs Ls, const bf16_t* __restrict__ dY, float* __restrict__ wsp,
    int B, int C, int ldy) {
  const int j = blockIdx.y;
  const bf16_t* L = Ls.p[j];
  const int ldl = Ls.ld[j];
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
    wsp[(int64_t)j * gridDim.x + blockIdx.x] =
        partial[0] + partial[1] + partial[2] + partial[3];
  }
}

__global__ __launch_bounds__(256) void mixer_bwd_dw_vector_ptrs_kernel(
    MixerPtrs Ls, const bf16_t* __restrict__ dY, float* __restrict__ dw,
    int B, int C, int ldy) {
  const int j = blockIdx.y;
  const bf16_t* L = Ls.p[j];
  const int ldl = Ls.ld[j];
  for (int c = blockIdx.x * blockDim.x + threadIdx.x; c < C;
       c += gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int b = 0; b < B; ++b)
      acc += bf2f(dY[(int64_t)b * ldy + c]) * bf2f(L[(int64_t)b * ldl + c]);
    // ACCUMULATE (dw pre-zeroed): direct-to-arena mode shares the grad
    // view with the L1-penalty term's AccumulateGrad add — overwrite
    // would clobber whichever landed first.
    dw[j * C + c] += acc;
  }
}

static int grid_for(int64_t total) {
  return (int)std::min<int64_t>((total + 255) / 256, 2048);
}

static MixerPtrs pack_ptrs(const std::vector<at::Tensor>& members, size_t j0,
                           int n, int B, int C) {
  MixerPtrs P;
  for (int k = 0; k < n; ++k) {
    const at::Tensor& m = members[j0 + k];
    TORCH_CHECK(m.is_cuda() && m.scalar_type() == at::kBFloat16 &&
                    m.dim() == 2 && (int)m.size(0) == B &&
                    (int)m.size(1) == C && m.stride(1) == 1,
                "mixer: member logits must be bf16 [B,C] row-strided");
    P.p[k] = (const bf16_t*)m.data_ptr();
    P.ld[k] = (int)m.stride(0);
  }
  for (int k = n; k < 8; ++k) {
    P.p[k] = nullptr;
    P.ld[k] = 0;
  }
  return P;
}

void mixer_fwd_direct(const std::vector<at::Tensor>& members,
                      const at::Tensor& weights,
                      const c10::optional<at::Tensor>& bias, at::Tensor& out,
                      int64_t vector_mode) {
  const int J = (int)members.size();
  TORCH_CHECK(J > 0, "mixer: need members");
  const int B = (int)members[0].size(0), C = (int)members[0].size(1);
  if ((int64_t)B * C == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  const float* bias_ptr =
      (bias.has_value() && bias->defined()) ? bias->data_ptr<float>() : nullptr;
  for (int j0 = 0; j0 < J; j0 += 8) {
    const int n = std::min(8, J - j0);
    MixerPtrs P = pack_ptrs(members, j0, n, B, C);
    const float* wp = weights.data_ptr<float>() +
                      (vector_mode ? (int64_t)j0 * C : (int64_t)j0);
    hipLaunchKernelGGL(mixer_fwd_ptrs_kernel,
                       dim3(grid_for((int64_t)B * C)), dim3(256), 0,
                       stream.stream(), P, wp, bias_ptr,
                       (bf16_t*)out.data_ptr(), n, B, C, (int)out.stride(0),
                       (int)vector_mode, j0 > 0 ? 1 : 0);
    HIP_CHECK_KERNEL();
  }
}

void mixer_bwd_dw_direct(const std::vector<at::Tensor>& members,
                         const at::Tensor& dY, at::Tensor& dw,
                         int64_t vector_mode) {
  const int J = (int)members.size();
  const int B = (int)members[0].size(0), C = (int)members[0].size(1);
  auto stream = at::cuda::getCurrentCUDAStream();
  for (int j0 = 0; j0 < J; j0 += 8) {
    const int n = std::min(8, J - j0);
    MixerPtrs P = pack_ptrs(members, j0, n, B, C);
    float* dwp = dw.data_ptr<float>() +
                 (vector_mode ? (int64_t)j0 * C : (int64_t)j0);
    if (vector_mode) {
      dim3 grid((unsigned)((C + 255) / 256), (unsigned)n);
      hipLaunchKernelGGL(mixer_bwd_dw_vector_ptrs_kernel, grid, dim3(256), 0,
                         stream.stream(), P, (const bf16_t*)dY.data_ptr(),
                         dwp, B, C, (int)dY.stride(0));
    } else {
      dim3 grid(
          (unsigned)std::min<int64_t>(((int64_t)B * C + 2047) / 2048, 256),
          (unsigned)n);
      auto wsp = at::empty({n, (int)grid.x}, dw.options());
      hipLaunchKernelGGL(mixer_bwd_dw_scalar_ptrs_kernel, grid, dim3(256), 0,
                         stream.stream(), P, (const bf16_t*)dY.data_ptr(),
                         wsp.data_ptr<float>(), B, C, (int)dY.stride(0));
      hipLaunchKernelGGL(mixer_dw_reduce_kernel, dim3(1), dim3(64), 0,
                         stream.stream(), wsp.data_ptr<float>(), dwp, n,
                         (int)grid.x);
    }
    HIP_CHECK_KERNEL();
  }
}

void mixer_fwd(const at::Tensor& stack, const at::Tensor& weights,
               const c10::optional<at::Tensor>& bias, at::Tensor& out,
               int64_t vector_mode) {
  TORCH_CHECK(stack.dim() == 3 && stack.is_contiguous(),
              "mixer: stacked [J,B,C] contiguous bf16 required");
  const int J = (int)stack.size(0), B = (int)stack.size(1),
            C = (int)stack.size(2);
  if (J == 0 || (int64_t)B * C == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  const float* bias_ptr =
      (bias.has_value() && bias->defined()) ? bias->data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(mixer_fwd_kernel, dim3(grid_for((int64_t)B * C)),
                     dim3(256), 0, stream.stream(),
                     (const bf16_t*)stack.data_ptr(),
                     weights.data_ptr<float>(), bias_ptr,
                     (bf16_t*)out.data_ptr(), J, B, C,
                     (int64_t)stack.stride(0), nullptr, (int)out.stride(0),
                     (int)vector_mode);
  HIP_CHECK_KERNEL();
}

void mixer_bwd_dw(const at::Tensor& stack, const at::Tensor& dY,
                  at::Tensor& dw, int64_t vector_mode) {
  const int J = (int)stack.size(0), B = (int)stack.size(1),
            C = (int)stack.size(2);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (vector_mode) {
    dim3 grid((unsigned)((C + 255) / 256), (unsigned)J);
    hipLaunchKernelGGL(mixer_bwd_dw_vector_kernel, grid, dim3(256), 0,
                       stream.stream(), (const bf16_t*)stack.data_ptr(),
                       (const bf16_t*)dY.data_ptr(), dw.data_ptr<float>(), J,
                       B, C, (int64_t)stack.stride(0), (int)dY.stride(0));
  } else {
    dim3 grid(
        (unsigned)std::min<int64_t>(((int64_t)B * C + 2047) / 2048, 256),
        (unsigned)J);
    auto wsp = at::empty({J, (int)grid.x}, dw.options());
    hipLaunchKernelGGL(mixer_bwd_dw_scalar_kernel, grid, dim3(256), 0,
                       stream.stream(), (const bf16_t*)stack.data_ptr(),
                       (const bf16_t*)dY.data_ptr(), wsp.data_ptr<float>(),
                       J, B, C, (int64_t)stack.stride(0), (int)dY.stride(0));
    hipLaunchKernelGGL(mixer_dw_reduce_kernel, dim3(1), dim3(64), 0,
                       stream.stream(), wsp.data_ptr<float>(),
                       dw.data_ptr<float>(), J, (int)grid.x);
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
