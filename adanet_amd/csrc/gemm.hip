// bf16 MFMA GEMM for gfx950 (MI355X): C[M,N] = A[M,K] * B[N,K]^T
// with optional fused epilogue (bias add + ReLU), fp32 accumulation.
//
// This is the K1 kernel of the AdaNet hot path (reference dependency:
// tf.layers.dense / tf.matmul in adanet/examples/simple_dnn.py:74-86 and
// the MATRIX mixture weights in adanet/ensemble/weighted.py:449) —
// re-designed CDNA4-native rather than ported:
//   * templated tile/wave-grid config (measured grid in
//     profiles/gemm_variants_r01*.json): 256x128 with 8 waves (64x64/wave)
//     for large problems, 128x128 with 16 waves (32x32/wave) for mid
//     shapes (half the LLC re-read traffic of a 64 tile at full occupancy
//     even at 1 block/CU), 64x64 for small/skinny shapes.
//   * BK=32 K-steps staged double-buffered with global_load_lds (16 B per
//     lane, wave-uniform LDS base: the direct HBM->LDS path).
//   * one s_barrier per K-tile; the compiler's vmcnt drain at the barrier
//     publishes the staged tile (the m97 structure from the CDNA4 guide).
//   * XCD-aware bijective blockIdx swizzle so neighboring output tiles
//     share a chiplet-local L2 (8 XCDs).
// All A/B/C layouts are row-major with the reduction dim (K) minor.
// Backward GEMMs consume K-major operands WITHOUT transposition via the
// ds_read_b64_tr_b16 transposed-staging kernels in gemm_tn.hip; skinny-M
// (M<=8, batch-1 serving) dispatches to a wave-per-column GEMV here.
// Epilogue act codes: 0 = none, 1 = ReLU, 2 = accumulate into C
// (direct-to-arena gradient writes).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

#define THREADS 256

// 8-phase deep-pipelined path (gemm_8ph.hip); returns false -> fall back.
bool gemm_nt8_try(const bf16_t* a, const bf16_t* b, bf16_t* c,
                  const float* bias_ptr, int M, int N, int K, int64_t lda,
                  int64_t ldb, int64_t ldc, int act, hipStream_t stream);
bool gemm_nt8_try_dropout(const bf16_t* a, const bf16_t* b, bf16_t* c,
                          const float* bias_ptr, int M, int N, int K,
                          int64_t lda, int64_t ldb, int64_t ldc,
                          const long long* seed, uint32_t pthresh,
                          float inv_keep, hipStream_t stream);

typedef s16x8 frag_ab;

// Stage a [ROWS x KSTEP] bf16 tile into LDS: each global_load_lds covers
// 1 KiB (64 lanes x 16 B, LDS-linear = 512/KSTEP rows); waves stride the
// segment list.
template <int ROWS, int KSTEP, int NWAVES = 4>
__device__ __forceinline__ void stage_tile_nt(
    const bf16_t* __restrict__ G, int ld, int tile_row0, int max_row, int k0,
    bf16_t* __restrict__ lds, int wid, int lane) {
  constexpr int ROWS_PER_SEG = 512 / KSTEP;      // rows per 1 KiB op
  constexpr int LANES_PER_ROW = KSTEP / 8;       // 16 B loads per row
  constexpr int SEGMENTS = ROWS / ROWS_PER_SEG;
#pragma unroll
  for (int seg = wid; seg < SEGMENTS; seg += NWAVES) {
    const int row_in_tile = seg * ROWS_PER_SEG + lane / LANES_PER_ROW;
    int grow = tile_row0 + row_in_tile;
    grow = grow < max_row ? grow : max_row - 1;  // clamp; masked on C-store
    const bf16_t* gp =
        G + (int64_t)grow * ld + k0 + (lane % LANES_PER_ROW) * 8;
    bf16_t* lp = lds + seg * ROWS_PER_SEG * KSTEP;  // wave-uniform base
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)gp,
        (__attribute__((address_space(3))) void*)lp, 16, 0, 0);
  }
}

// BM x BN tile, 2x2 wave grid, FM x FN 16x16 MFMA fragments per wave,
// KSTEP K-columns staged per barrier (KSTEP/32 MFMA K-slices).
// SPLITK > 1: the grid carries SPLITK K-slabs per output tile; each slab
// writes its fp32 partial into its OWN C32 slot [split][M][N] (no bias/
// act), and a separate epilogue kernel sums the slots in fixed order and
// finishes bias+ReLU+bf16 — deterministic (fp32 atomic ordering used to
// wobble conv dW), no zero-fill, no atomic contention; trades a little
// output traffic for filling all 256 CUs with the efficient big tile.
template <int BM, int BN, int FM, int FN, int MINWAVES, int KSTEP = 32,
          bool SPLITK = false, int WGM = 2, int WGN = 2, int GROUPM = 0>
__global__ __launch_bounds__(WGM * WGN * 64, MINWAVES) void gemm_nt_bf16_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, int lda, int ldb, int ldc, int act, int mtiles, int ntiles,
    float* __restrict__ C32 = nullptr, int ksplit = 1,
    const long long* __restrict__ seed = nullptr, uint32_t pthresh = 0,
    float inv_keep = 1.f) {
  __shared__ bf16_t As[2][BM * KSTEP];
  __shared__ bf16_t Bs[2][BN * KSTEP];

  // Bijective XCD-aware swizzle (guide m204): contiguous tile chunks/XCD.
  const int nwg = mtiles * ntiles * (SPLITK ? ksplit : 1);
  const int orig = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7, xcd = orig & 7;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) +
                 (orig >> 3);
  const int tiles = mtiles * ntiles;
  const int split = SPLITK ? (wg / tiles) : 0;
  const int tile_lin = SPLITK ? (wg % tiles) : wg;
  // GROUPM > 0: L2 supertile grouping — consecutive tiles walk M within a
  // GROUPM-tall column slab, so each XCD's contiguous chunk touches a
  // square-ish C region and re-reads A/B slabs from its own L2 instead of
  // streaming all of B per M-band (row-major order). Bijective incl. the
  // tail band (last band's height gm < GROUPM).
  int tile_m, tile_n;
  if (GROUPM > 0) {
    const int per_band = GROUPM * ntiles;
    const int band = tile_lin / per_band;
    const int in_band = tile_lin % per_band;
    const int gm = min(GROUPM, mtiles - band * GROUPM);
    tile_m = band * GROUPM + in_band % gm;
    tile_n = in_band / gm;
  } else {
    tile_m = tile_lin / ntiles;
    tile_n = tile_lin % ntiles;
  }

  static_assert(BM == WGM * FM * 16 && BN == WGN * FN * 16,
                "tile must equal wave grid x fragments");
  constexpr int NWAVES = WGM * WGN;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int wm = wid / WGN, wn = wid % WGN;  // WGM x WGN wave grid

  const int row0 = tile_m * BM;
  const int col0 = tile_n * BN;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  int kt0 = 0, ktend = K / KSTEP;
  if (SPLITK) {
    const int per = (ktend + ksplit - 1) / ksplit;
    kt0 = split * per;
    ktend = min(ktend, kt0 + per);
    if (kt0 >= ktend) return;
  }
  stage_tile_nt<BM, KSTEP, NWAVES>(A, lda, row0, M, kt0 * KSTEP, As[0], wid,
                                   lane);
  stage_tile_nt<BN, KSTEP, NWAVES>(B, ldb, col0, N, kt0 * KSTEP, Bs[0], wid,
                                   lane);

  int buf = 0;
  for (int kt = kt0; kt < ktend; ++kt) {
    // Explicit per-wave drain of the LDS-DMA issued last iteration:
    // __syncthreads() alone does not reliably order global_load_lds
    // completion against cross-wave LDS reads (the tr path reads via
    // inline-asm ds_read, invisible to the compiler's waitcnt insertion
    // -- observed as a run-to-run race in the NASNet 1x1 conv batched
    // GEMMs, benchmarks/nas_det_probe.py).
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();  // staged tile `buf` visible; prior reads of buf^1 done
    if (kt + 1 < ktend) {
      const int k0 = (kt + 1) * KSTEP;
      stage_tile_nt<BM, KSTEP, NWAVES>(A, lda, row0, M, k0, As[buf ^ 1],
                                       wid, lane);
      stage_tile_nt<BN, KSTEP, NWAVES>(B, ldb, col0, N, k0, Bs[buf ^ 1],
                                       wid, lane);
    }
#pragma unroll
    for (int kk = 0; kk < KSTEP / 32; ++kk) {
      // Fragment loads: 8 contiguous bf16 per lane -> ds_read_b128.
      frag_ab a[FM], b[FN];
      const int kofs = kk * 32 + (lane >> 4) * 8;
      const int arow = wm * (FM * 16) + (lane & 15);
      const int brow = wn * (FN * 16) + (lane & 15);
#pragma unroll
      for (int f = 0; f < FM; ++f)
        a[f] = *(const frag_ab*)&As[buf][(arow + f * 16) * KSTEP + kofs];
#pragma unroll
      for (int f = 0; f < FN; ++f)
        b[f] = *(const frag_ab*)&Bs[buf][(brow + f * 16) * KSTEP + kofs];
#pragma unroll
      for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    buf ^= 1;
  }

  // Epilogue: C/D layout (16x16x32): col = lane&15, row = (lane>>4)*4 + reg.
  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < FM; ++i) {
#pragma unroll
    for (int j = 0; j < FN; ++j) {
      const int col = col0 + wn * (FN * 16) + j * 16 + c_col_in_frag;
      if (col >= N) continue;
      const float bv = (!SPLITK && bias) ? bias[col] : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int row = row0 + wm * (FM * 16) + i * 16 + c_row_base + rr;
        if (row >= M) continue;
        if (SPLITK) {
          C32[((int64_t)split * M + row) * N + col] = acc[i][j][rr];
        } else {
          float v = acc[i][j][rr] + bv;
          if (act == 1) v = v > 0.f ? v : 0.f;
          if (act == 3) {  // fused relu + dropout (stateless counter RNG)
            v = v > 0.f ? v : 0.f;
            const uint32_t r =
                hash_rng((uint64_t)seed[0], (uint64_t)row * N + col);
            v = (r >= pthresh) ? v * inv_keep : 0.f;
          }
          // act==2: accumulate into C (direct-to-arena gradient writes).
          bf16_t* cp = &C[(int64_t)row * ldc + col];
          if (act == 2) v += bf2f(*cp);
          *cp = f2bf(v);
        }
      }
    }
  }
}

// Finishes a split-K accumulation: bf16 C = act(sum_slots C32 + bias),
// summing the ksplit slots in fixed order (deterministic).
__global__ __launch_bounds__(256) void gemm_splitk_epilogue_kernel(
    const float* __restrict__ C32, const float* __restrict__ bias,
    bf16_t* __restrict__ C, int M, int N, int ldc, int act, int ksplit) {
  const int64_t total = (int64_t)M * N;
  for (int64_t p = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; p < total;
       p += (int64_t)gridDim.x * blockDim.x) {
    const int m = (int)(p / N), n = (int)(p % N);
    float v = bias ? bias[n] : 0.f;
    for (int r = 0; r < ksplit; ++r) v += C32[(int64_t)r * total + p];
    if (act == 1) v = v > 0.f ? v : 0.f;
    if (act == 2) v += bf2f(C[(int64_t)m * ldc + n]);
    C[(int64_t)m * ldc + n] = f2bf(v);
  }
}

// High-ksplit epilogue twin: one WAVE per output element, lanes over
// the ksplit slots, fixed-tree wave reduce (deterministic). The
// thread-per-element kernel serialized ksplit loads on a near-empty
// chip for the skinny conv-dW shapes (M*N ~5k, ksplit ~150: 30 us
// where traffic is ~1 us — 4.4% of the NASNet step,
// profiles/nasprof7_summary.txt).
__global__ __launch_bounds__(256) void gemm_splitk_epilogue_wave_kernel(
    const float* __restrict__ C32, const float* __restrict__ bias,
    bf16_t* __restrict__ C, int M, int N, int ldc, int act, int ksplit) {
  const int64_t total = (int64_t)M * N;
  const int64_t p = (int64_t)blockIdx.x * 4 + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  if (p >= total) return;
  float v = 0.f;
  for (int r = lane; r < ksplit; r += 64) v += C32[(int64_t)r * total + p];
  v = wave_reduce_sum(v);
  if (lane == 0) {
    const int m = (int)(p / N), n = (int)(p % N);
    v += bias ? bias[n] : 0.f;
    if (act == 1) v = v > 0.f ? v : 0.f;
    if (act == 2) v += bf2f(C[(int64_t)m * ldc + n]);
    C[(int64_t)m * ldc + n] = f2bf(v);
  }
}

// GEMV path for skinny-M GEMMs (M <= 8: batch-1/small-batch serving,
// where an MFMA tile would waste 63/64 rows): one wave per output column,
// lanes stride K with 16 B loads, wave-shuffle reduction. Bandwidth-bound
// on the B matrix read (~10x the tiled kernel's batch-1 latency).
struct bfx8 {
  bf16_t v[8];
};

__global__ __launch_bounds__(256) void gemm_nt_gemv_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, int lda, int ldb, int ldc, int act,
    const long long* __restrict__ seed = nullptr, uint32_t pthresh = 0,
    float inv_keep = 1.f) {
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int n = blockIdx.x * 4 + wid;
  if (n >= N) return;
  const bf16_t* brow = B + (int64_t)n * ldb;
  float acc[8];
#pragma unroll
  for (int m = 0; m < 8; ++m) acc[m] = 0.f;
  for (int k = lane * 8; k < K; k += 64 * 8) {
    const bfx8 bv = *(const bfx8*)(brow + k);
    for (int m = 0; m < M; ++m) {
      const bfx8 av = *(const bfx8*)(A + (int64_t)m * lda + k);
#pragma unroll
      for (int i = 0; i < 8; ++i) acc[m] += bf2f(av.v[i]) * bf2f(bv.v[i]);
    }
  }
  for (int m = 0; m < M; ++m) {
    float v = wave_reduce_sum(acc[m]);
    if (lane == 0) {
      if (bias) v += bias[n];
      if (act == 1) v = v > 0.f ? v : 0.f;
      if (act == 3) {
        v = v > 0.f ? v : 0.f;
        const uint32_t r =
            hash_rng((uint64_t)seed[0], (uint64_t)m * N + n);
        v = (r >= pthresh) ? v * inv_keep : 0.f;
      }
      bf16_t* cp = &C[(int64_t)m * ldc + n];
      if (act == 2) v += bf2f(*cp);
      *cp = f2bf(v);
    }
  }
}

// Skinny-N path (classifier logits: N=10, M=batch, K=hidden). An MFMA
// 64-wide tile wastes (64-N)/64 of its work AND fills only M/64 blocks
// (measured 33 us / 2.4 TF on the bench logits GEMM, 6.7% of the step,
// profiles/bench_kernel_stats_r02d.txt). Here the WHOLE B matrix (N*K
// bf16 <= 64 KB) is staged in LDS once per block; each wave then streams
// rows of A with 16 B loads and keeps all N dot products in registers —
// VALU-bound but ~10x faster than the wasted MFMA tile at these shapes.
__global__ __launch_bounds__(256) void gemm_nt_skinnyn_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, int lda, int ldb, int ldc, int act) {
  __shared__ bf16_t bsmem[32768];  // 64 KB: N * K <= 32768 (host gate)
  for (int n = 0; n < N; ++n) {
    const bf16_t* src = B + (int64_t)n * ldb;
    bf16_t* dst = bsmem + n * K;
    for (int i = threadIdx.x * 8; i < K; i += 256 * 8)
      *(bfx8*)(dst + i) = *(const bfx8*)(src + i);
  }
  __syncthreads();
  const int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
  for (int m = blockIdx.x * 4 + wid; m < M; m += gridDim.x * 4) {
    float acc[16];
#pragma unroll
    for (int n = 0; n < 16; ++n) acc[n] = 0.f;
    const bf16_t* arow = A + (int64_t)m * lda;
    for (int k = lane * 8; k < K; k += 64 * 8) {
      const bfx8 av = *(const bfx8*)(arow + k);
      float af[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) af[i] = bf2f(av.v[i]);
      for (int n = 0; n < N; ++n) {
        const bfx8 bv = *(const bfx8*)(bsmem + n * K + k);
#pragma unroll
        for (int i = 0; i < 8; ++i) acc[n] += af[i] * bf2f(bv.v[i]);
      }
    }
    for (int n = 0; n < N; ++n) {
      float v = wave_reduce_sum(acc[n]);
      if (lane == 0) {
        if (bias) v += bias[n];
        if (act == 1) v = v > 0.f ? v : 0.f;
        bf16_t* cp = &C[(int64_t)m * ldc + n];
        if (act == 2) v += bf2f(*cp);
        *cp = f2bf(v);
      }
    }
  }
}

// Generic any-stride fallback (correctness net for shapes the fast path
// can't take: lda/ldb not 8-aligned or K not a multiple of 32). VALU fp32.
__global__ void gemm_nt_generic_kernel(const bf16_t* __restrict__ A,
                                       const bf16_t* __restrict__ B,
                                       bf16_t* __restrict__ C,
                                       const float* __restrict__ bias, int M,
                                       int N, int K, int lda, int ldb, int ldc,
                                       int act,
                                       const long long* __restrict__ seed
                                       = nullptr,
                                       uint32_t pthresh = 0,
                                       float inv_keep = 1.f) {
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t total = (int64_t)M * N;
  for (int64_t p = idx; p < total; p += (int64_t)gridDim.x * blockDim.x) {
    const int m = (int)(p / N), n = (int)(p % N);
    float acc = 0.f;
    for (int k = 0; k < K; ++k)
      acc += bf2f(A[(int64_t)m * lda + k]) * bf2f(B[(int64_t)n * ldb + k]);
    if (bias) acc += bias[n];
    if (act == 1) acc = acc > 0.f ? acc : 0.f;
    if (act == 3) {
      acc = acc > 0.f ? acc : 0.f;
      const uint32_t r = hash_rng((uint64_t)seed[0], (uint64_t)m * N + n);
      acc = (r >= pthresh) ? acc * inv_keep : 0.f;
    }
    if (act == 2) acc += bf2f(C[(int64_t)m * ldc + n]);
    C[(int64_t)m * ldc + n] = f2bf(acc);
  }
}

void gemm_nt_bf16(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
                  const c10::optional<at::Tensor>& bias, int64_t act,
                  double dropout_p, const c10::optional<at::Tensor>& seed) {
  TORCH_CHECK(A.is_cuda() && B.is_cuda() && C.is_cuda(), "gemm: need GPU tensors");
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 && B.scalar_type() == at::kBFloat16,
              "gemm: bf16 inputs required");
  TORCH_CHECK(A.dim() == 2 && B.dim() == 2 && C.dim() == 2, "gemm: 2-D only");
  TORCH_CHECK(A.stride(1) == 1 && B.stride(1) == 1 && C.stride(1) == 1,
              "gemm: innermost dim must be contiguous");
  const int M = (int)A.size(0), K = (int)A.size(1), N = (int)B.size(0);
  TORCH_CHECK(B.size(1) == K, "gemm: K mismatch");
  TORCH_CHECK(C.size(0) == M && C.size(1) == N, "gemm: C shape mismatch");
  const int lda = (int)A.stride(0), ldb = (int)B.stride(0),
            ldc = (int)C.stride(0);
  if (M == 0 || N == 0) return;
  if (K == 0) { C.zero_(); return; }
  const float* bias_ptr = nullptr;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->scalar_type() == at::kFloat && bias->numel() == N,
                "gemm: bias must be fp32[N]");
    bias_ptr = bias->data_ptr<float>();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  const bf16_t* a = (const bf16_t*)A.data_ptr();
  const bf16_t* b = (const bf16_t*)B.data_ptr();
  bf16_t* c = (bf16_t*)C.data_ptr();

  // act==3: fused relu+dropout epilogue (stateless counter RNG keyed on a
  // DEVICE seed snapshot -> hipGraph replays draw fresh masks).
  const long long* seed_ptr = nullptr;
  uint32_t pthresh = 0;
  float inv_keep = 1.f;
  if (act == 3) {
    TORCH_CHECK(seed.has_value() && seed->defined() && seed->is_cuda() &&
                seed->scalar_type() == at::kLong,
                "gemm: act=3 needs a device int64 seed tensor");
    TORCH_CHECK(dropout_p > 0.0 && dropout_p < 1.0, "gemm: bad dropout_p");
    seed_ptr = (const long long*)seed->data_ptr();
    pthresh = (uint32_t)(dropout_p * 4294967296.0);
    inv_keep = (float)(1.0 / (1.0 - dropout_p));
  }

  const bool fast = (K % 32 == 0) && (lda % 8 == 0) && (ldb % 8 == 0);
  if (fast && M <= 8) {
    hipLaunchKernelGGL(gemm_nt_gemv_kernel, dim3((N + 3) / 4), dim3(256), 0,
                       stream.stream(), a, b, c, bias_ptr, M, N, K, lda, ldb,
                       ldc, (int)act, seed_ptr, pthresh, inv_keep);
    HIP_CHECK_KERNEL();
    return;
  }
  if (fast && act != 3 && N <= 16 && (int64_t)N * K <= 32768 && M >= 64) {
    hipLaunchKernelGGL(gemm_nt_skinnyn_kernel,
                       dim3(std::min((M + 3) / 4, 2048)), dim3(256), 0,
                       stream.stream(), a, b, c, bias_ptr, M, N, K, lda, ldb,
                       ldc, (int)act);
    HIP_CHECK_KERNEL();
    return;
  }
  // K-dominant tiny-output GEMMs (conv-space pointwise dW: M=N=channels,
  // K=B*H*W up to ~262k): the tile grid is 1-8 workgroups, so the plain
  // kernel runs one CU for a ms-scale reduction (measured 49% of the
  // improve_nas step, profiles/nasprof_summary.txt). Split the K range
  // over ~512 workgroups with slotted fp32 partials + a tiny epilogue.
  const int64_t t64 = (int64_t)((M + 63) / 64) * ((N + 63) / 64);
  const bool k_dominant =
      (t64 <= 8 && K >= 2048) ||
      (t64 <= 256 && K >= 4096 && (int64_t)K >= 4 * std::max(M, N));
  if (fast && act != 3 && k_dominant &&
      std::min<int64_t>(512 / t64, (int64_t)K / 256) >= 2) {
    int ksplit = (int)std::min<int64_t>(512 / t64, (int64_t)K / 256);
    if (ksplit < 2) ksplit = 2;
    // Clamp so every split owns >=1 K-tile: an empty split would leave its
    // C32 slot UNWRITTEN and the epilogue would sum garbage (observed as
    // the NASNet 1x1 dW nondeterminism, benchmarks/nas_det_probe.py).
    {
      const int ktiles = K / 32;
      const int per = (ktiles + ksplit - 1) / ksplit;
      ksplit = (ktiles + per - 1) / per;
      if (ksplit < 2) ksplit = 2;
    }
    auto C32 = at::empty({ksplit, M, N}, A.options().dtype(at::kFloat));
    const int mt = (M + 63) / 64, nt = (N + 63) / 64;
    hipLaunchKernelGGL((gemm_nt_bf16_kernel<64, 64, 2, 2, 6, 32, true>),
                       dim3(mt * nt * ksplit), dim3(THREADS), 0,
                       stream.stream(), a, b, c, nullptr, M, N, K, lda, ldb,
                       ldc, (int)act, mt, nt, C32.data_ptr<float>(), ksplit);
    const int64_t tot = (int64_t)M * N;
    if (ksplit > 16) {
      hipLaunchKernelGGL(gemm_splitk_epilogue_wave_kernel,
                         dim3((unsigned)((tot + 3) / 4)), dim3(256), 0,
                         stream.stream(), C32.data_ptr<float>(), bias_ptr, c,
                         M, N, ldc, (int)act, ksplit);
    } else {
      hipLaunchKernelGGL(gemm_splitk_epilogue_kernel,
                         dim3((int)std::min<int64_t>((tot + 255) / 256,
                                                     2048)),
                         dim3(256), 0, stream.stream(),
                         C32.data_ptr<float>(), bias_ptr, c, M, N, ldc,
                         (int)act, ksplit);
    }
    HIP_CHECK_KERNEL();
    return;
  }
  if (fast && act != 3 &&
      gemm_nt8_try(a, b, c, bias_ptr, M, N, K, lda, ldb, ldc,
                   (int)act, stream.stream())) {
    HIP_CHECK_KERNEL();
    return;
  }
  if (fast && act == 3 &&
      gemm_nt8_try_dropout(a, b, c, bias_ptr, M, N, K, lda, ldb, ldc,
                           seed_ptr, pthresh, inv_keep, stream.stream())) {
    HIP_CHECK_KERNEL();
    return;
  }
  if (fast) {
    // Tile dispatch tuned on MI355X (benchmarks/gemm_variants.py,
    // profiles/gemm_variants_r01.json): occupancy rules until ~4 blocks/CU,
    // then per-wave efficiency of the big tile wins; for asymmetric mid
    // shapes put the 128 side on the SHORTER output dim.
    const int mt128 = (M + 127) / 128, nt128 = (N + 127) / 128;
    const int64_t b128 = (int64_t)mt128 * nt128;
#define LAUNCH_CFG(BM, BN, FM, FN, MW)                                        \
  do {                                                                        \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    hipLaunchKernelGGL((gemm_nt_bf16_kernel<BM, BN, FM, FN, MW, 32>),         \
                       dim3(mt * nt), dim3(THREADS), 0, stream.stream(), a,  \
                       b, c, bias_ptr, M, N, K, lda, ldb, ldc, (int)act, mt, \
                       nt, nullptr, 1, seed_ptr, pthresh, inv_keep);          \
  } while (0)
#define LAUNCH_CFG_W(BM, BN, FM, FN, MW, WGM, WGN, GM)                        \
  do {                                                                        \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    hipLaunchKernelGGL(                                                       \
        (gemm_nt_bf16_kernel<BM, BN, FM, FN, MW, 32, false, WGM, WGN, GM>),   \
        dim3(mt * nt), dim3(WGM * WGN * 64), 0, stream.stream(), a, b, c,    \
        bias_ptr, M, N, K, lda, ldb, ldc, (int)act, mt, nt, nullptr, 1,       \
        seed_ptr, pthresh, inv_keep);                                         \
  } while (0)
    const int64_t t_256x128 =
        (int64_t)((M + 255) / 256) * ((N + 127) / 128);
    const int64_t t_128x256 =
        (int64_t)((M + 127) / 128) * ((N + 255) / 256);
    if (M >= N && t_256x128 >= 512) {
      // 256x128, 8 waves (4x2), 64x64/wave: big-tile L2/LLC traffic with
      // 2 waves/SIMD (980 TF @4096^3, profiles/gemm_variants_r01c).
      LAUNCH_CFG_W(256, 128, 4, 4, 2, 4, 2, 0);
    } else if (N > M && t_128x256 >= 512) {
      LAUNCH_CFG_W(128, 256, 4, 4, 2, 2, 4, 0);
    } else if (b128 >= 160) {
      // 128^2, 16 waves (4x4), 32x32/wave, G8 L2 supertile grouping:
      // half the 64^2 tile's LLC re-reads at full occupancy; grouping
      // +21% at 4096-class shapes that miss the 256-tile threshold,
      // neutral at 2048-class (gemm_group_probe_r01.json). A 64^2
      // K-dominant carve-out was measured BOTH ways across boxes
      // (±5% box noise, gemm_mid_probe) — dropped for stability.
      LAUNCH_CFG_W(128, 128, 2, 2, 4, 4, 4, 8);
    } else {
      LAUNCH_CFG(64, 64, 2, 2, 6);
    }
#undef LAUNCH_CFG
#undef LAUNCH_CFG_W
  } else {
    const int64_t total = (int64_t)M * N;
    const int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
    hipLaunchKernelGGL(gemm_nt_generic_kernel, dim3(blocks), dim3(256), 0,
                       stream.stream(), a, b, c, bias_ptr, M, N, K, lda, ldb,
                       ldc, (int)act, seed_ptr, pthresh, inv_keep);
  }
  HIP_CHECK_KERNEL();
}

// Tuning probe: dispatch a specific tile/occupancy/K-step variant so a
// single GPU session can A/B the whole configuration grid
// (benchmarks/gemm_variants.py).
void gemm_nt_bf16_probe(const at::Tensor& A, const at::Tensor& B,
                        at::Tensor& C, int64_t variant) {
  const int M = (int)A.size(0), K = (int)A.size(1), N = (int)B.size(0);
  const int lda = (int)A.stride(0), ldb = (int)B.stride(0),
            ldc = (int)C.stride(0);
  auto stream = at::cuda::getCurrentCUDAStream();
  const bf16_t* a = (const bf16_t*)A.data_ptr();
  const bf16_t* b = (const bf16_t*)B.data_ptr();
  bf16_t* c = (bf16_t*)C.data_ptr();

#define LAUNCH_V(BM, BN, FM, FN, MW, KS)                                      \
  do {                                                                        \
    TORCH_CHECK(K % KS == 0, "probe: K %% KSTEP");                            \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    hipLaunchKernelGGL((gemm_nt_bf16_kernel<BM, BN, FM, FN, MW, KS>),         \
                       dim3(mt * nt), dim3(THREADS), 0, stream.stream(), a,  \
                       b, c, nullptr, M, N, K, lda, ldb, ldc, 0, mt, nt);     \
  } while (0)

#define LAUNCH_SK(BM, BN, FM, FN, MW, KS, SPLIT)                              \
  do {                                                                        \
    TORCH_CHECK(K % KS == 0, "probe: K %% KSTEP");                            \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    auto C32 = at::empty({(int64_t)SPLIT, (int64_t)M, (int64_t)N},            \
                         A.options().dtype(at::kFloat));                      \
    hipLaunchKernelGGL((gemm_nt_bf16_kernel<BM, BN, FM, FN, MW, KS, true>),   \
                       dim3(mt * nt * SPLIT), dim3(THREADS), 0,              \
                       stream.stream(), a, b, c, nullptr, M, N, K, lda, ldb, \
                       ldc, 0, mt, nt, C32.data_ptr<float>(), SPLIT);         \
    const int64_t tot = (int64_t)M * N;                                       \
    hipLaunchKernelGGL(gemm_splitk_epilogue_kernel,                           \
                       dim3((int)std::min<int64_t>((tot + 255) / 256, 2048)), \
                       dim3(256), 0, stream.stream(),                         \
                       C32.data_ptr<float>(), nullptr, c, M, N, ldc, 0,      \
                       SPLIT);                                                \
  } while (0)

  switch (variant) {
    case 0: LAUNCH_V(128, 128, 4, 4, 2, 32); break;
    case 1: LAUNCH_V(128, 128, 4, 4, 3, 32); break;
    case 2: LAUNCH_V(128, 128, 4, 4, 4, 32); break;
    case 3: LAUNCH_V(64, 64, 2, 2, 4, 32); break;
    case 4: LAUNCH_V(64, 64, 2, 2, 8, 32); break;
    case 5: LAUNCH_V(128, 64, 4, 2, 4, 32); break;
    case 6: LAUNCH_V(128, 64, 4, 2, 2, 32); break;
    case 7: LAUNCH_V(128, 128, 4, 4, 2, 64); break;
    case 8: LAUNCH_V(128, 64, 4, 2, 4, 64); break;
    case 9: LAUNCH_V(64, 64, 2, 2, 6, 32); break;
    case 10: LAUNCH_V(64, 128, 2, 4, 4, 32); break;
    case 11: LAUNCH_SK(128, 128, 4, 4, 2, 32, 2); break;
    case 12: LAUNCH_SK(128, 128, 4, 4, 2, 32, 4); break;
    case 13: LAUNCH_SK(128, 128, 4, 4, 4, 32, 4); break;
    case 14: LAUNCH_SK(128, 128, 4, 4, 4, 32, 2); break;
    case 15: LAUNCH_SK(64, 64, 2, 2, 6, 32, 2); break;
    case 16: LAUNCH_V(64, 64, 2, 2, 6, 64); break;
#define LAUNCH_VW(BM, BN, FM, FN, MW, WGM, WGN)                               \
  do {                                                                        \
    TORCH_CHECK(K % 32 == 0, "probe: K %% 32");                               \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    hipLaunchKernelGGL(                                                       \
        (gemm_nt_bf16_kernel<BM, BN, FM, FN, MW, 32, false, WGM, WGN>),       \
        dim3(mt * nt), dim3(WGM * WGN * 64), 0, stream.stream(), a, b, c,    \
        nullptr, M, N, K, lda, ldb, ldc, 0, mt, nt);                          \
  } while (0)
    case 17: LAUNCH_V(64, 64, 2, 2, 4, 64); break;
    case 18: LAUNCH_V(64, 128, 2, 4, 4, 64); break;
    case 19: LAUNCH_V(64, 128, 2, 4, 2, 64); break;
    case 20: LAUNCH_VW(128, 128, 2, 2, 2, 4, 4); break;   // 16 waves
    case 21: LAUNCH_VW(128, 128, 4, 2, 2, 2, 4); break;   // 8 waves 64x32
    case 22: LAUNCH_VW(128, 128, 2, 4, 2, 4, 2); break;   // 8 waves 32x64
    case 23: LAUNCH_VW(256, 128, 4, 4, 2, 4, 2); break;   // 8 waves 64x64
    case 24: LAUNCH_VW(128, 256, 4, 4, 2, 2, 4); break;
    case 25: LAUNCH_VW(128, 128, 2, 2, 4, 4, 4); break;   // 16 waves mw4
    case 26: LAUNCH_VW(256, 128, 4, 2, 2, 4, 4); break;   // 16w, 64x32/wave
    case 27: LAUNCH_VW(256, 256, 4, 4, 2, 4, 4); break;   // 16w, 64x64/wave
    case 28: LAUNCH_VW(256, 64, 4, 2, 2, 4, 2); break;    // 8w, 64x32/wave
    case 29: {  // 256x128 8w with KSTEP=64
      TORCH_CHECK(K % 64 == 0, "probe: K %% 64");
      const int mt = (M + 255) / 256, nt = (N + 127) / 128;
      hipLaunchKernelGGL(
          (gemm_nt_bf16_kernel<256, 128, 4, 4, 2, 64, false, 4, 2>),
          dim3(mt * nt), dim3(512), 0, stream.stream(), a, b, c, nullptr, M,
          N, K, lda, ldb, ldc, 0, mt, nt);
      break;
    }
    case 30: {  // 128x128 16w with KSTEP=64
      TORCH_CHECK(K % 64 == 0, "probe: K %% 64");
      const int mt = (M + 127) / 128, nt = (N + 127) / 128;
      hipLaunchKernelGGL(
          (gemm_nt_bf16_kernel<128, 128, 2, 2, 4, 64, false, 4, 4>),
          dim3(mt * nt), dim3(1024), 0, stream.stream(), a, b, c, nullptr, M,
          N, K, lda, ldb, ldc, 0, mt, nt);
      break;
    }
#define LAUNCH_VG(BM, BN, FM, FN, MW, WGM, WGN, GM)                           \
  do {                                                                        \
    TORCH_CHECK(K % 32 == 0, "probe: K %% 32");                               \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    hipLaunchKernelGGL(                                                       \
        (gemm_nt_bf16_kernel<BM, BN, FM, FN, MW, 32, false, WGM, WGN, GM>),   \
        dim3(mt * nt), dim3(WGM * WGN * 64), 0, stream.stream(), a, b, c,    \
        nullptr, M, N, K, lda, ldb, ldc, 0, mt, nt);                          \
  } while (0)
    // L2 supertile grouping sweep (GROUPM column-slab order)
    case 31: LAUNCH_VG(128, 128, 2, 2, 4, 4, 4, 2); break;
    case 32: LAUNCH_VG(128, 128, 2, 2, 4, 4, 4, 4); break;
    case 33: LAUNCH_VG(128, 128, 2, 2, 4, 4, 4, 8); break;
    case 34: LAUNCH_VG(128, 128, 2, 2, 4, 4, 4, 16); break;
    case 35: LAUNCH_VG(256, 128, 4, 4, 2, 4, 2, 2); break;
    case 36: LAUNCH_VG(256, 128, 4, 4, 2, 4, 2, 4); break;
    case 37: LAUNCH_VG(64, 64, 2, 2, 6, 2, 2, 4); break;
    case 38: LAUNCH_VG(64, 64, 2, 2, 6, 2, 2, 8); break;
    case 39: LAUNCH_VG(128, 256, 4, 4, 2, 2, 4, 2); break;
    case 40: LAUNCH_VG(128, 256, 4, 4, 2, 2, 4, 4); break;
    // mid-tile alternatives for the 2048-class bench shapes
    case 41: LAUNCH_VG(96, 96, 3, 3, 4, 2, 2, 0); break;
    case 42: LAUNCH_VG(96, 96, 3, 3, 4, 2, 2, 8); break;
    case 43: LAUNCH_VG(64, 128, 2, 4, 4, 2, 2, 0); break;
    case 44: LAUNCH_VG(128, 64, 4, 2, 4, 2, 2, 0); break;
    case 45: LAUNCH_VG(64, 128, 2, 2, 4, 2, 4, 0); break;   // 8w 32x32/wave
    case 46: LAUNCH_VG(96, 96, 3, 3, 2, 2, 2, 0); break;    // minwaves 2
    case 47: LAUNCH_VG(64, 64, 2, 2, 4, 2, 2, 0); break;    // 64^2 mw4
    case 48: LAUNCH_VG(64, 64, 2, 2, 8, 2, 2, 0); break;    // 64^2 mw8
    default: TORCH_CHECK(false, "unknown variant");
  }
#undef LAUNCH_V
#undef LAUNCH_SK
  HIP_CHECK_KERNEL();
}
