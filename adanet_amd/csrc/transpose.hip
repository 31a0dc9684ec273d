// LDS-tiled bf16 2-D transpose: out[N,M] = in[M,N]^T.
//
// Used by the linear-layer backward to put both GEMM operands in K-minor
// layout for gemm.hip (dX needs W^T, dW needs dY^T and X^T). Memory-bound:
// 64x64 tiles staged through LDS with +8-halfword row padding so the
// strided read side avoids bank conflicts; ushort4 (8 B) vector loads on
// the coalesced side (guide G13: never scalar bf16).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

#define TDIM 64
#define TPAD 8  // pad leading dim: 64+8 halfwords -> read stride 144 B, bank-spread

typedef __attribute__((ext_vector_type(4))) unsigned short us16x4;

__global__ __launch_bounds__(256) void transpose_bf16_kernel(
    const bf16_t* __restrict__ in, bf16_t* __restrict__ out, int M, int N,
    int ldi, int ldo, int mtiles, int ntiles) {
  __shared__ bf16_t tile[TDIM][TDIM + TPAD];
  const int tm = blockIdx.x / ntiles;  // tile row (over M)
  const int tn = blockIdx.x % ntiles;  // tile col (over N)
  const int t = threadIdx.x;
  const int r4 = t >> 4;        // 0..15
  const int c4 = (t & 15) * 4;  // column group of 4
  // Interior tiles take the fully-vectorized path (ushort4 both sides).
  const bool interior = (tm * TDIM + TDIM <= M) && (tn * TDIM + TDIM <= N) &&
                        (ldi % 4 == 0) && (ldo % 4 == 0);

  // Load: rows of `in` coalesced.
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int row = p * 16 + r4;
    const int gr = tm * TDIM + row;
    const int gc = tn * TDIM + c4;
    if (interior) {
      const us16x4 v =
          *(const us16x4*)&in[(int64_t)gr * ldi + gc];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const unsigned short s = v[u];
        tile[row][c4 + u] = *(const bf16_t*)&s;
      }
    } else if (gr < M) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int c = gc + u;
        tile[row][c4 + u] = (c < N) ? in[(int64_t)gr * ldi + c] : f2bf(0.f);
      }
    }
  }
  __syncthreads();
  // Store: rows of `out` (columns of `in`) coalesced, packed ushort4.
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int row = p * 16 + r4;       // row of out == col of in
    const int gr = tn * TDIM + row;
    const int gc = tm * TDIM + c4;
    if (interior) {
      us16x4 v;
#pragma unroll
      for (int u = 0; u < 4; ++u)
        v[u] = *(const unsigned short*)&tile[c4 + u][row];
      *(us16x4*)&out[(int64_t)gr * ldo + gc] = v;
    } else if (gr < N) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        const int c = gc + u;
        if (c < M) out[(int64_t)gr * ldo + c] = tile[c4 + u][row];
      }
    }
  }
}

void transpose_bf16(const at::Tensor& in, at::Tensor& out) {
  TORCH_CHECK(in.is_cuda() && out.is_cuda(), "transpose: need GPU tensors");
  TORCH_CHECK(in.scalar_type() == at::kBFloat16, "transpose: bf16 only");
  TORCH_CHECK(in.dim() == 2 && out.dim() == 2, "transpose: 2-D only");
  TORCH_CHECK(in.stride(1) == 1 && out.stride(1) == 1, "transpose: row-major");
  const int M = (int)in.size(0), N = (int)in.size(1);
  TORCH_CHECK(out.size(0) == N && out.size(1) == M, "transpose: shape");
  if (M == 0 || N == 0) return;
  const int mtiles = (M + TDIM - 1) / TDIM, ntiles = (N + TDIM - 1) / TDIM;
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(transpose_bf16_kernel, dim3(mtiles * ntiles), dim3(256),
                     0, stream.stream(), (const bf16_t*)in.data_ptr(),
                     (bf16_t*)out.data_ptr(), M, N, (int)in.stride(0),
                     (int)out.stride(0), mtiles, ntiles);
  HIP_CHECK_KERNEL();
}
