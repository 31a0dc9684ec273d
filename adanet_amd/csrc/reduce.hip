// Reduction kernels (K6/K10 support): column-sum of a bf16 matrix into fp32
// (bias gradients: db[c] = sum_b dY[b,c]) and streaming metric reductions
// (correct-prediction count for accuracy, reference eval_metrics.py usage).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

// 2-D grid: (column stripes, row chunks). Each block reduces its row chunk
// for a 256-column stripe (coalesced: consecutive threads read consecutive
// columns of each row); the row-chunk axis is what fills the 256-CU chip
// (a single-stripe launch occupied 8 CUs and was 54% of the training step
// before this). Multi-chunk partials land in `wsp` and a fixed-order
// reducer finishes — bit-deterministic across runs.
__global__ __launch_bounds__(256) void colsum_bf16_kernel(
    const bf16_t* __restrict__ x, float* __restrict__ out, int B, int C,
    int ldx, int rows_per_block, int accum, float* __restrict__ wsp) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const int r0 = blockIdx.y * rows_per_block;
  const int r1 = min(B, r0 + rows_per_block);
  float acc = 0.f;
  for (int b = r0; b < r1; ++b) acc += bf2f(x[(int64_t)b * ldx + c]);
  if (gridDim.y == 1) {
    if (accum) out[c] += acc;  // single chunk: deterministic add
    else out[c] = acc;
  } else {
    // multi-chunk: per-chunk workspace slot; a fixed-order reducer
    // finishes (atomics had run-dependent fp32 ordering).
    wsp[(int64_t)blockIdx.y * C + c] = acc;
  }
}

// out[c] (+)= sum_chunk wsp[chunk][c]: one wave per column, lanes over
// chunks, fixed-tree wave reduce — deterministic and chip-filling (the
// thread-per-column version serialized nchunks loads; at C=10 it ran 10
// threads x 256 loads and was 10% of the DNN step).
__global__ __launch_bounds__(256) void colsum_reduce_kernel(
    const float* __restrict__ wsp, float* __restrict__ out, int C,
    int nchunks, int accum) {
  const int c = blockIdx.x * 4 + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (c >= C) return;
  float s = 0.f;
  for (int r = lane; r < nchunks; r += 64) s += wsp[(int64_t)r * C + c];
  s = wave_reduce_sum(s);
  if (lane == 0) {
    if (accum) out[c] += s;
    else out[c] = s;
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

// Score histograms for streaming binary-classification metrics (K10 tail:
// AUC / precision / recall, reference tf.metrics.auc's thresholded
// TP/FP/TN/FN accumulation). Buckets scores in [0,1] into T bins, one
// histogram per class, LDS-accumulated then merged with one atomic per
// bin per block. Suffix sums over the histograms give TP/FP at every
// threshold (core/eval_metrics.py _AUCAccumulator).
__global__ __launch_bounds__(256) void binary_histogram_kernel(
    const float* __restrict__ scores, const int64_t* __restrict__ labels,
    int* __restrict__ hist, int B, int T) {
  extern __shared__ int lh[];  // [2][T]
  for (int i = threadIdx.x; i < 2 * T; i += blockDim.x) lh[i] = 0;
  __syncthreads();
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < B;
       i += stride) {
    float s = scores[i];
    s = s < 0.f ? 0.f : (s > 1.f ? 1.f : s);
    int b = (int)(s * T);
    b = b >= T ? T - 1 : b;
    atomicAdd(&lh[(labels[i] != 0 ? 0 : 1) * T + b], 1);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < 2 * T; i += blockDim.x) {
    if (lh[i]) atomicAdd(&hist[i], lh[i]);
  }
}

void binary_histogram(const at::Tensor& scores, const at::Tensor& labels,
                      at::Tensor& hist) {
  TORCH_CHECK(scores.scalar_type() == at::kFloat && scores.is_contiguous(),
              "binary_histogram: contiguous fp32 scores");
  TORCH_CHECK(labels.scalar_type() == at::kLong, "binary_histogram: int64 labels");
  TORCH_CHECK(hist.scalar_type() == at::kInt && hist.dim() == 2 &&
                  hist.size(0) == 2,
              "binary_histogram: int32 [2,T] hist");
  const int B = (int)scores.numel(), T = (int)hist.size(1);
  if (B == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  const int blocks = std::min((B + 255) / 256, 512);
  hipLaunchKernelGGL(binary_histogram_kernel, dim3(blocks), dim3(256),
                     2 * T * sizeof(int), stream.stream(),
                     scores.data_ptr<float>(), labels.data_ptr<int64_t>(),
                     hist.data_ptr<int>(), B, T);
  HIP_CHECK_KERNEL();
}

void colsum_bf16(const at::Tensor& x, at::Tensor& out, int64_t accum) {
  const int B = (int)x.size(0), C = (int)x.size(1);
  if (C == 0) return;
  if (B == 0) { if (!accum) out.zero_(); return; }
  auto stream = at::cuda::getCurrentCUDAStream();
  const int stripes = (C + 255) / 256;
  // Fill the chip: aim for ~1024 blocks, at least 8 rows per chunk.
  int row_chunks = std::max(1, std::min(1024 / stripes, (B + 7) / 8));
  const int rows_per_block = (B + row_chunks - 1) / row_chunks;
  if (row_chunks > 1) {
    auto wsp = at::empty({row_chunks, C}, out.options());
    hipLaunchKernelGGL(colsum_bf16_kernel,
                       dim3((unsigned)stripes, (unsigned)row_chunks),
                       dim3(256), 0, stream.stream(),
                       (const bf16_t*)x.data_ptr(), out.data_ptr<float>(), B,
                       C, (int)x.stride(0), rows_per_block, (int)accum,
                       wsp.data_ptr<float>());
    hipLaunchKernelGGL(colsum_reduce_kernel, dim3((unsigned)((C + 3) / 4)),
                       dim3(256), 0, stream.stream(),
                       wsp.data_ptr<float>(), out.data_ptr<float>(), C,
                       row_chunks, (int)accum);
    HIP_CHECK_KERNEL();
    return;
  }
  hipLaunchKernelGGL(colsum_bf16_kernel,
                     dim3((unsigned)stripes, (unsigned)row_chunks), dim3(256),
                     0, stream.stream(), (const bf16_t*)x.data_ptr(),
                     out.data_ptr<float>(), B, C, (int)x.stride(0),
                     rows_per_block, (int)accum, (float*)nullptr);
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
