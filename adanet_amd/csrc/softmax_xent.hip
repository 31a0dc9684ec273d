// Fused softmax-cross-entropy head (K2): forward computes per-row loss and
// softmax probabilities in one pass; backward is the closed-form
// dlogits = (p - target) * grad_row, with optional label smoothing
// (target = (1-eps)*onehot + eps/C, matching the reference's
// tf.losses.softmax_cross_entropy(label_smoothing=...) used by
// research/improve_nas/trainer/improve_nas.py:160-181 and the head call in
// adanet/core/ensemble_builder.py:571-583).
//
// Layout: logits bf16 [B, C] with explicit row stride (the engine pads the
// logits dimension to 8-element alignment for the GEMM fast path and hands
// this kernel the narrow view). One 64-lane wave per row; lanes stride C;
// wave shuffle reductions for max and sum-exp (no LDS needed).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

__global__ __launch_bounds__(256) void softmax_xent_fwd_kernel(
    const bf16_t* __restrict__ logits, const int64_t* __restrict__ labels,
    float* __restrict__ loss, bf16_t* __restrict__ probs, int B, int C,
    int ldl, int ldp, float eps, float* __restrict__ mean_out,
    float* __restrict__ partials) {
  const int wave_in_block = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int waves = (gridDim.x * blockDim.x) >> 6;
  float block_sum = 0.f;  // per-wave running sum for the fused mean
  for (int row = blockIdx.x * (blockDim.x >> 6) + wave_in_block; row < B;
       row += waves) {
    const bf16_t* lrow = logits + (int64_t)row * ldl;
    float mx = -INFINITY;
    for (int c = lane; c < C; c += 64) mx = fmaxf(mx, bf2f(lrow[c]));
    mx = wave_reduce_max(mx);
    float se = 0.f;
    for (int c = lane; c < C; c += 64) se += __expf(bf2f(lrow[c]) - mx);
    se = wave_reduce_sum(se);
    const float lse = __logf(se) + mx;
    const int64_t y = labels[row];
    // loss = lse - (1-eps)*logit_y - (eps/C)*sum(logits)
    float sum_logits = 0.f;
    float ly = 0.f;
    const float inv_se = 1.f / se;
    for (int c = lane; c < C; c += 64) {
      const float v = bf2f(lrow[c]);
      sum_logits += v;
      if (c == (int)y) ly = v;
      if (probs) probs[(int64_t)row * ldp + c] = f2bf(__expf(v - mx) * inv_se);
    }
    sum_logits = wave_reduce_sum(sum_logits);
    ly = wave_reduce_sum(ly);  // only the label lane contributed
    if (lane == 0) {
      const float l = lse - (1.f - eps) * ly - (eps / C) * sum_logits;
      if (loss) loss[row] = l;
      block_sum += l;
    }
  }
  // Fused mean: per-block LDS reduction, then either a plain per-block
  // partial store (preferred: no pre-zero fill needed, a tiny reducer
  // kernel finishes) or ONE global atomic per block (legacy path, needs a
  // zeroed mean_out; a per-row atomic to one address serialized ~6x).
  if (mean_out || partials) {
    __shared__ float partial[4];
    if (lane == 0) partial[wave_in_block] = block_sum;
    __syncthreads();
    if (threadIdx.x == 0) {
      float s = 0.f;
      const int nwaves = blockDim.x >> 6;
      for (int w = 0; w < nwaves; ++w) s += partial[w];
      if (partials) {
        partials[blockIdx.x] = s;  // overwrite: scratch never zeroed
      } else {
        atomicAdd(mean_out, s / B);
      }
    }
  }
}

// Finishes the fused mean: mean = sum(partials) / B. One wave, overwrite
// store — the output scalar needs no pre-zeroing (removes one fill kernel
// per loss call from every captured train step).
__global__ __launch_bounds__(64) void xent_mean_reduce_kernel(
    const float* __restrict__ partials, int n, float inv_b,
    float* __restrict__ out) {
  float s = 0.f;
  for (int i = threadIdx.x; i < n; i += 64) s += partials[i];
  s = wave_reduce_sum(s);
  if (threadIdx.x == 0) out[0] = s * inv_b;
}

__global__ __launch_bounds__(256) void softmax_xent_bwd_kernel(
    const bf16_t* __restrict__ probs, const int64_t* __restrict__ labels,
    const float* __restrict__ grad_rows, bf16_t* __restrict__ dlogits, int B,
    int C, int ldp, int ldd, float eps,
    const float* __restrict__ grad_scalar) {
  const int wave_in_block = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int waves = (gridDim.x * blockDim.x) >> 6;
  // grad_scalar: upstream grad of the MEAN loss (device pointer, read once
  // per row) -> per-row grad = *grad_scalar / B. Else per-row grad_rows.
  for (int row = blockIdx.x * (blockDim.x >> 6) + wave_in_block; row < B;
       row += waves) {
    const int64_t y = labels[row];
    const float g = grad_scalar ? (*grad_scalar) / B : grad_rows[row];
    for (int c = lane; c < C; c += 64) {
      float t = (c == (int)y ? 1.f - eps : 0.f) + eps / C;
      float p = bf2f(probs[(int64_t)row * ldp + c]);
      dlogits[(int64_t)row * ldd + c] = f2bf((p - t) * g);
    }
  }
}

void softmax_xent_fwd(const at::Tensor& logits, const at::Tensor& labels,
                      const c10::optional<at::Tensor>& loss,
                      at::Tensor& probs, double eps,
                      const c10::optional<at::Tensor>& mean_out) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == at::kBFloat16,
              "xent: bf16 GPU logits required");
  TORCH_CHECK(labels.scalar_type() == at::kLong, "xent: int64 labels");
  TORCH_CHECK(logits.stride(1) == 1, "xent: contiguous class dim");
  const int B = (int)logits.size(0), C = (int)logits.size(1);
  if (B == 0 || C == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  const bool fused_mean = mean_out.has_value() && mean_out->defined();
  // With the fused mean, cap blocks so the final atomics stay few; each
  // wave loops multiple rows instead.
  const int blocks = std::min((B + 3) / 4, fused_mean ? 256 : 2048);
  float* loss_ptr = (loss.has_value() && loss->defined())
                        ? loss->data_ptr<float>()
                        : nullptr;
  float* mean_ptr = (mean_out.has_value() && mean_out->defined())
                        ? mean_out->data_ptr<float>()
                        : nullptr;
  // Fused mean via per-block partials + one-wave reducer: mean_out is
  // overwritten, never pre-zeroed (no fill kernel per loss call).
  at::Tensor scratch;
  float* partials_ptr = nullptr;
  if (fused_mean) {
    scratch = at::empty({blocks}, logits.options().dtype(at::kFloat));
    partials_ptr = scratch.data_ptr<float>();
  }
  hipLaunchKernelGGL(softmax_xent_fwd_kernel, dim3(blocks), dim3(256), 0,
                     stream.stream(), (const bf16_t*)logits.data_ptr(),
                     labels.data_ptr<int64_t>(), loss_ptr,
                     (bf16_t*)probs.data_ptr(), B, C, (int)logits.stride(0),
                     (int)probs.stride(0), (float)eps,
                     partials_ptr ? nullptr : mean_ptr, partials_ptr);
  HIP_CHECK_KERNEL();
  if (fused_mean) {
    hipLaunchKernelGGL(xent_mean_reduce_kernel, dim3(1), dim3(64), 0,
                       stream.stream(), partials_ptr, blocks, 1.f / B,
                       mean_ptr);
    HIP_CHECK_KERNEL();
  }
}

void softmax_xent_bwd(const at::Tensor& probs, const at::Tensor& labels,
                      const c10::optional<at::Tensor>& grad_rows,
                      at::Tensor& dlogits, double eps,
                      const c10::optional<at::Tensor>& grad_scalar) {
  const int B = (int)probs.size(0), C = (int)probs.size(1);
  if (B == 0 || C == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  const int blocks = std::min((B + 3) / 4, 2048);
  const float* gr = (grad_rows.has_value() && grad_rows->defined())
                        ? grad_rows->data_ptr<float>()
                        : nullptr;
  const float* gs = (grad_scalar.has_value() && grad_scalar->defined())
                        ? grad_scalar->data_ptr<float>()
                        : nullptr;
  TORCH_CHECK(gr || gs, "xent bwd: need grad_rows or grad_scalar");
  hipLaunchKernelGGL(softmax_xent_bwd_kernel, dim3(blocks), dim3(256), 0,
                     stream.stream(), (const bf16_t*)probs.data_ptr(),
                     labels.data_ptr<int64_t>(), gr,
                     (bf16_t*)dlogits.data_ptr(), B, C, (int)probs.stride(0),
                     (int)dlogits.stride(0), (float)eps, gs);
  HIP_CHECK_KERNEL();
}
