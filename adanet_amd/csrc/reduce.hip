// Reduction kernels (K6/K10 support): column-sum of a bf16 matrix into fp32
// (bias gradients: db[c] = sum_b dY[b,c]) and streaming metric reductions
// (correct-prediction count for accuracy, reference eval_metrics.py usage).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

// 2-D grid: (column stripes, row chunks). Each block reduces its row chunk
// for a 256-column stripe (coalesced: consecutive threads read consecutive
// columns of each row) and atomically adds one partial per column — the
// row-chunk axis is what fills the 256-CU chip (a single-stripe launch
// occupied 8 CUs and was 54% of the training step before this).
// `out` MUST be zero-initialized when gridDim.y > 1.
__global__ __launch_bounds__(256) void colsum_bf16_kernel(
    const bf16_t* __restrict__ x, float* __restrict__ out, int B, int C,
    int ldx, int rows_per_block, int accum) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const int r0 = blockIdx.y * rows_per_block;
  const int r1 = min(B, r0 + rows_per_block);
  float acc = 0.f;
  for (int b = r0; b < r1; ++b) acc += bf2f(x[(int64_t)b * ldx + c]);
  if (gridDim.y == 1 && !accum) {
    out[c] = acc;
  } else {
    // accum mode: out already holds the running gradient (direct-to-arena
    // bias-grad write) — always add, never overwrite, never pre-zero.
    atomicAdd(&out[c], acc);
  }
}

// argmax over the class dim + count of matches with labels (accuracy numer).
__global__ __launch_bounds__(256) void argmax_correct_kernel(
    const bf16_t* __restrict__ logits, const int64_t* __restrict__ labels,
    int64_t* __restrict__ pred, int* __restrict__ correct, int B, int C,
    int ldl) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += stride) {
    const bf16_t* row = logits + b * ldl;
    float best = bf2f(row[0]);
    int arg = 0;
    for (int c = 1; c < C; ++c) {
      const float v = bf2f(row[c]);
      if (v > best) {
        best = v;
        arg = c;
      }
    }
    if (pred) pred[b] = arg;
    if (correct && arg == (int)labels[b]) atomicAdd(correct, 1);
  }
}

void colsum_bf16(const at::Tensor& x, at::Tensor& out, int64_t accum) {
  const int B = (int)x.size(0), C = (int)x.size(1);
  if (C == 0) return;
  if (B == 0) { if (!accum) out.zero_(); return; }
  auto stream = at::cuda::getCurrentCUDAStream();
  const int stripes = (C + 255) / 256;
  // Fill the chip: aim for ~1024 blocks, at least 8 rows per chunk.
  int row_chunks = std::max(1, std::min(1024 / stripes, (B + 7) / 8));
  if (row_chunks > 1 && !accum) {
    out.zero_();
  }
  const int rows_per_block = (B + row_chunks - 1) / row_chunks;
  hipLaunchKernelGGL(colsum_bf16_kernel,
                     dim3((unsigned)stripes, (unsigned)row_chunks), dim3(256),
                     0, stream.stream(), (const bf16_t*)x.data_ptr(),
                     out.data_ptr<float>(), B, C, (int)x.stride(0),
                     rows_per_block, (int)accum);
  HIP_CHECK_KERNEL();
}

void argmax_correct(const at::Tensor& logits, const at::Tensor& labels,
                    const c10::optional<at::Tensor>& pred,
                    at::Tensor& correct) {
  const int B = (int)logits.size(0), C = (int)logits.size(1);
  if (B == 0 || C == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  int64_t* pred_ptr =
      (pred && pred->defined()) ? pred->data_ptr<int64_t>() : nullptr;
  hipLaunchKernelGGL(argmax_correct_kernel,
                     dim3(std::min((B + 255) / 256, 2048)), dim3(256), 0,
                     stream.stream(), (const bf16_t*)logits.data_ptr(),
                     labels.data_ptr<int64_t>(), pred_ptr,
                     correct.data_ptr<int>(), B, C, (int)logits.stride(0));
  HIP_CHECK_KERNEL();
}
