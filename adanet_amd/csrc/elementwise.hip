// Elementwise kernels (K3): dropout fwd/bwd with a stateless counter-based
// RNG (mask recomputed in backward from (seed, index) — nothing stored),
// ReLU backward (mask recovered from the saved fused-forward OUTPUT, so the
// GEMM epilogue never materializes a mask), and bf16 cast helpers.
// All vectorized 8-wide bf16 via short4 pairs where shapes allow
// (guide G13: scalar bf16 loads are 2-2.5x slower).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

// The seed arrives via DEVICE memory (seed_ptr): a per-module device
// counter advances each forward, so the op is hipGraph-capturable with a
// fresh mask per replay (a host-scalar seed would bake one mask into the
// graph). Backward reads the forward's snapshot tensor.
__global__ __launch_bounds__(256) void dropout_fwd_kernel(
    const bf16_t* __restrict__ x, bf16_t* __restrict__ y, int64_t n, float p,
    const uint64_t* __restrict__ seed_ptr) {
  const float scale = 1.f / (1.f - p);
  const uint32_t thresh = (uint32_t)(p * 4294967296.0f);
  const uint64_t seed = *seed_ptr;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const bool drop = hash_rng(seed, (uint64_t)i) < thresh;
    y[i] = drop ? f2bf(0.f) : f2bf(bf2f(x[i]) * scale);
  }
}

__global__ __launch_bounds__(256) void dropout_bwd_kernel(
    const bf16_t* __restrict__ dy, bf16_t* __restrict__ dx, int64_t n, float p,
    const uint64_t* __restrict__ seed_ptr) {
  const float scale = 1.f / (1.f - p);
  const uint32_t thresh = (uint32_t)(p * 4294967296.0f);
  const uint64_t seed = *seed_ptr;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const bool drop = hash_rng(seed, (uint64_t)i) < thresh;
    dx[i] = drop ? f2bf(0.f) : f2bf(bf2f(dy[i]) * scale);
  }
}

// dX = dY * (Y > 0): Y is the fused GEMM+ReLU output.
__global__ __launch_bounds__(256) void relu_bwd_kernel(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ y,
    bf16_t* __restrict__ dx, int64_t n, float scale) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    // scale = 1/(1-p) under the fused relu+dropout epilogue: y>0 identifies
    // kept-and-positive elements, so dropout backward is a scaled mask.
    dx[i] = bf2f(y[i]) > 0.f ? f2bf(bf2f(dy[i]) * scale) : f2bf(0.f);
  }
}

// Fused dz = dY * (Y > 0) AND db += colsum(dz): the bias gradient is
// accumulated while the masked gradient is produced, saving the separate
// colsum kernel's full re-read of dz (8 MB at the bench shape; colsum was
// 3.7% + relu_bwd 3.0% of step GPU time, profiles/bench_kernel_stats_r01e).
// Layout mirrors colsum: thread-per-column (coalesced), row-chunk grid.y
// fills the chip (direct-to-arena accum semantics — db never pre-zeroed).
// Deterministic: each (stripe, chunk) block writes its partial into its
// OWN workspace slot; a fixed-order reducer finishes (fp32 atomics have
// run-dependent ordering -> last-ulp bias-grad wobble that flips near-tie
// candidate selections across otherwise-identical runs).
__global__ __launch_bounds__(256) void relu_bwd_colsum_kernel(
    const bf16_t* __restrict__ dy, const bf16_t* __restrict__ y,
    bf16_t* __restrict__ dz, float* __restrict__ wsp, int B, int C,
    int rows_per_block, float scale) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const int r0 = blockIdx.y * rows_per_block;
  const int r1 = min(B, r0 + rows_per_block);
  float acc = 0.f;
  for (int b = r0; b < r1; ++b) {
    const int64_t i = (int64_t)b * C + c;
    const float g = bf2f(y[i]) > 0.f ? bf2f(dy[i]) * scale : 0.f;
    dz[i] = f2bf(g);
    acc += g;
  }
  wsp[(int64_t)blockIdx.y * C + c] = acc;
}

// db[c] += sum_chunk wsp[chunk][c], chunks reduced lane-parallel within
// one wave per column then wave-shuffled in FIXED TREE ORDER (still
// deterministic). Thread-per-column serialized nchunks loads on a
// near-empty chip (was 11% of the DNN step, bench_kernel_stats_r02c).
__global__ __launch_bounds__(256) void colsum_chunk_reduce_kernel(
    const float* __restrict__ wsp, float* __restrict__ db, int C,
    int nchunks) {
  const int c = blockIdx.x * 4 + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (c >= C) return;
  float s = 0.f;
  for (int r = lane; r < nchunks; r += 64) s += wsp[(int64_t)r * C + c];
  s = wave_reduce_sum(s);
  if (lane == 0) db[c] += s;
}

// Batched device-to-device copy: up to 8 (src,dst) pairs per launch
// (frozen-logit cache -> static graph-input buffers each step was J-1
// separate ~7.6 us copyBuffer launches; one kernel replaces them).
struct CopyPtrs {
  const bf16_t* s[8];
  bf16_t* d[8];
  int64_t n[8];
};

typedef __attribute__((ext_vector_type(4))) unsigned int u32x4_c;

__global__ __launch_bounds__(256) void multi_copy_bf16_kernel(CopyPtrs P,
                                                              int npairs) {
  const int seg = blockIdx.y;
  if (seg >= npairs) return;
  const bf16_t* __restrict__ s = P.s[seg];
  bf16_t* __restrict__ d = P.d[seg];
  const int64_t n = P.n[seg];
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       i < n; i += stride) {
    if (i + 8 <= n) {
      *(u32x4_c*)(d + i) = *(const u32x4_c*)(s + i);
    } else {
      for (int64_t j = i; j < n; ++j) d[j] = s[j];
    }
  }
}

void multi_copy_bf16(const std::vector<at::Tensor>& srcs,
                     const std::vector<at::Tensor>& dsts) {
  TORCH_CHECK(srcs.size() == dsts.size(), "multi_copy: pair count mismatch");
  auto stream = at::cuda::getCurrentCUDAStream();
  for (size_t j0 = 0; j0 < srcs.size(); j0 += 8) {
    const int np = (int)std::min<size_t>(8, srcs.size() - j0);
    CopyPtrs P;
    int64_t maxn = 0;
    for (int k = 0; k < np; ++k) {
      const at::Tensor& s = srcs[j0 + k];
      const at::Tensor& d = dsts[j0 + k];
      TORCH_CHECK(s.is_cuda() && d.is_cuda() &&
                      s.scalar_type() == at::kBFloat16 &&
                      d.scalar_type() == at::kBFloat16 &&
                      s.is_contiguous() && d.is_contiguous() &&
                      s.numel() == d.numel(),
                  "multi_copy: contiguous bf16 same-size pairs required");
      P.s[k] = (const bf16_t*)s.data_ptr();
      P.d[k] = (bf16_t*)d.data_ptr();
      P.n[k] = s.numel();
      maxn = std::max(maxn, P.n[k]);
    }
    if (maxn == 0) continue;
    const int blocks =
        (int)std::min<int64_t>((maxn + 8 * 256 - 1) / (8 * 256), 512);
    hipLaunchKernelGGL(multi_copy_bf16_kernel,
                       dim3((unsigned)blocks, (unsigned)np), dim3(256), 0,
                       stream.stream(), P, np);
    HIP_CHECK_KERNEL();
  }
}

static int ew_grid(int64_t n) {
  return (int)std::min<int64_t>((n + 255) / 256, 2048);
}

void dropout_fwd(const at::Tensor& x, at::Tensor& y, double p,
                 const at::Tensor& seed) {
  const int64_t n = x.numel();
  if (n == 0) return;
  TORCH_CHECK(seed.is_cuda() && seed.scalar_type() == at::kLong,
              "dropout: device int64 seed tensor required");
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(dropout_fwd_kernel, dim3(ew_grid(n)), dim3(256), 0,
                     stream.stream(), (const bf16_t*)x.data_ptr(),
                     (bf16_t*)y.data_ptr(), n, (float)p,
                     (const uint64_t*)seed.data_ptr());
  HIP_CHECK_KERNEL();
}

void dropout_bwd(const at::Tensor& dy, at::Tensor& dx, double p,
                 const at::Tensor& seed) {
  const int64_t n = dy.numel();
  if (n == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(dropout_bwd_kernel, dim3(ew_grid(n)), dim3(256), 0,
                     stream.stream(), (const bf16_t*)dy.data_ptr(),
                     (bf16_t*)dx.data_ptr(), n, (float)p,
                     (const uint64_t*)seed.data_ptr());
  HIP_CHECK_KERNEL();
}

void relu_bwd(const at::Tensor& dy, const at::Tensor& y, at::Tensor& dx,
              double scale) {
  const int64_t n = dy.numel();
  if (n == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(relu_bwd_kernel, dim3(ew_grid(n)), dim3(256), 0,
                     stream.stream(), (const bf16_t*)dy.data_ptr(),
                     (const bf16_t*)y.data_ptr(), (bf16_t*)dx.data_ptr(), n,
                     (float)scale);
  HIP_CHECK_KERNEL();
}

void relu_bwd_colsum(const at::Tensor& dy, const at::Tensor& y,
                     at::Tensor& dz, at::Tensor& db, double scale) {
  TORCH_CHECK(dy.is_contiguous() && y.is_contiguous() && dz.is_contiguous(),
              "relu_bwd_colsum: contiguous tensors required");
  TORCH_CHECK(db.scalar_type() == at::kFloat, "relu_bwd_colsum: fp32 db");
  const int B = (int)dy.size(0), C = (int)dy.size(1);
  if (B == 0 || C == 0) return;
  auto stream = at::cuda::getCurrentCUDAStream();
  const int stripes = (C + 255) / 256;
  const int row_chunks =
      std::max(1, std::min(1024 / stripes, (B + 7) / 8));
  const int rows_per_block = (B + row_chunks - 1) / row_chunks;
  auto wsp = at::empty({row_chunks, C}, db.options());
  hipLaunchKernelGGL(relu_bwd_colsum_kernel,
                     dim3((unsigned)stripes, (unsigned)row_chunks), dim3(256),
                     0, stream.stream(), (const bf16_t*)dy.data_ptr(),
                     (const bf16_t*)y.data_ptr(), (bf16_t*)dz.data_ptr(),
                     wsp.data_ptr<float>(), B, C, rows_per_block,
                     (float)scale);
  hipLaunchKernelGGL(colsum_chunk_reduce_kernel,
                     dim3((unsigned)((C + 3) / 4)), dim3(256), 0,
                     stream.stream(), wsp.data_ptr<float>(),
                     db.data_ptr<float>(), C, row_chunks);
  HIP_CHECK_KERNEL();
}
