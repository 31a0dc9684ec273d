// 8-phase deep-pipelined bf16 MFMA GEMM for gfx950: C[M,N] = A[M,K]*B[N,K]^T.
//
// The round-1 kernel (gemm.hip) uses the 2-barrier-per-K-step structure whose
// measured ceiling is ~900 TF: __syncthreads() drains vmcnt(0) so the staged
// tile pipeline empties at every barrier. This kernel implements the CDNA4
// guide's 256^2 8-phase schedule (measured 1563 TF @4096^3 in the guide's
// m201) re-derived for this codebase:
//
//   * BK=64 K-tiles, double-buffered; each tile split into 4 half-tiles
//     (A rows [0,BM/2), [BM/2,BM); B likewise) staged by block-wide
//     global_load_lds rounds (512 threads x 16 B = 64 rows of 128 B/round).
//   * 4 compute phases per K-tile; phase p computes m-reps {p*FM/4..} x all
//     n-reps x 2 k-steps. B fragments are ds_read ONCE per K-tile (phase 0)
//     and held in VGPRs; A fragments are read per phase.
//   * raw s_barrier (NOT __syncthreads) so the global_load_lds queue is
//     never drained at barriers; an explicit counted s_waitcnt vmcnt(N)
//     runs ONCE per K-tile (phase 3), leaving the two youngest half-tile
//     prefetches in flight across the tile boundary.
//   * prefetch schedule (provable invariant, see below): during tile t,
//     phases 0/1 issue A-halves of tile t+1 (other LDS buffer); phases 2/3
//     issue B-halves of tile t+2 (SAME buffer as t: B slots are dead after
//     t's phase 0 because B fragments live in registers).
//       issue order: ... B0(t) B1(t) A0(t) A1(t) B0(t+1) B1(t+1) ...
//       at tile t-1 phase 3: s_waitcnt vmcnt(2*LB) leaves exactly the
//       B(t+1) pair outstanding => every half of tile t has landed, and
//       each wave's wait + the phase barrier publishes all waves' staging.
//   * LDS st-swizzle: 16-byte block index XOR (row&7) kills the 128 B
//     row-stride bank conflict on fragment ds_read_b128 (16-way -> 2-way;
//     2-way is free on CDNA4). global_load_lds writes LDS linearly, so the
//     swizzle is applied by permuting each lane's GLOBAL source block
//     (involution) and reading LDS with the same XOR.
//   * s_setprio(1) around the MFMA cluster (guide T5: +21-25% on this
//     structure), XCD-aware bijective tile swizzle (T1).
//
// Reference dependency (K1): tf.layers.dense/tf.matmul hot path,
// adanet/examples/simple_dnn.py:74-86. SAFE=true template variant keeps
// __syncthreads-style drains for A/B correctness bisection.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace {

// Stage one half-tile (HROWS x 64 bf16) into LDS with the block's 8 waves.
// Linear LDS dest (global_load_lds constraint); swizzled global source.
// HROWS in {64, 128} -> 1 or 2 rounds; each lane issues HROWS/64 loads.
template <int HROWS>
__device__ __forceinline__ void stage_half(
    const bf16_t* __restrict__ G, int64_t ld, int rows0, int max_row, int k0,
    bf16_t* __restrict__ lds_half, int wid, int lane) {
#pragma unroll
  for (int r = 0; r < HROWS / 64; ++r) {
    const int row_in_half = r * 64 + wid * 8 + (lane >> 3);
    int grow = rows0 + row_in_half;
    grow = grow < max_row ? grow : max_row - 1;  // clamp; masked on C-store
    const int src_blk = (lane & 7) ^ ((lane >> 3) & 7);  // st-swizzle
    const bf16_t* gp = G + (int64_t)grow * ld + k0 + src_blk * 8;
    bf16_t* lp = lds_half + (r * 64 + wid * 8) * 64;  // wave-uniform base
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)gp,
        (__attribute__((address_space(3))) void*)lp, 16, 0, 0);
  }
}

// Swizzled LDS fragment read: logical (row, 8-element block) of a [ROWS][64]
// bf16 tile whose 16 B blocks were XOR-permuted by (row&7) at staging.
//
// The read is INLINE ASM on purpose: a C++ LDS load after a pending
// global_load_lds makes the compiler insert s_waitcnt vmcnt(0) before it
// (it cannot prove the in-flight DMA writes a different LDS region), which
// drains the prefetch pipeline at every phase — measured 927 vs the
// counted-vmcnt schedule this kernel is built around. The schedule's own
// vmcnt/barrier invariant is what guarantees the data has landed. Callers
// MUST wait lgkmcnt(0) + sched_barrier(0) before consuming the result
// (asm ds_read data arrives asynchronously; the compiler doesn't know).
__device__ __forceinline__ s16x8 lds_frag(const bf16_t* lds, int row,
                                          int blk8) {
  const int pblk = blk8 ^ (row & 7);
  const uint32_t addr = (uint32_t)(uintptr_t)(
      const __attribute__((address_space(3))) void*)&lds[row * 64 + pblk * 8];
  s16x8 out;
  asm volatile("ds_read_b128 %0, %1" : "=v"(out) : "v"(addr));
  return out;
}

#define S_BARRIER() __builtin_amdgcn_s_barrier()

// s_waitcnt vmcnt(N) with a compile-time literal (asm strings can't expand
// constexpr values; dispatch over the small set of counts this kernel uses).
template <int N>
__device__ __forceinline__ void vmcnt_wait() {
  static_assert((N >= 0 && N <= 4) || N == 6 || N == 8,
                "unsupported vmcnt");
  if constexpr (N == 6) {
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    return;
  }
  if constexpr (N == 8) {
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    return;
  }
  if constexpr (N == 0) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  } else if constexpr (N == 1) {
    asm volatile("s_waitcnt vmcnt(1)" ::: "memory");
  } else if constexpr (N == 2) {
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
  } else if constexpr (N == 3) {
    asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
  } else if constexpr (N == 4) {
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  }
}

// BM x BN output tile, 8 waves (2 M x 4 N), BK=64, 4 phases per K-tile.
// SAFE: drain-everything barriers (correctness bisection baseline).
template <int BM, int BN, bool SAFE = false, bool SETPRIO = true>
__global__ __launch_bounds__(512, 2) void gemm_nt8_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, int64_t lda, int64_t ldb, int64_t ldc, int act, int mtiles,
    int ntiles, const long long* __restrict__ seed = nullptr,
    uint32_t pthresh = 0, float inv_keep = 1.f) {
  constexpr int FM = BM / (2 * 16);   // m-fragments per wave
  constexpr int FN = BN / (4 * 16);   // n-fragments per wave
  constexpr int MG = FM / 4;          // m-reps per phase
  static_assert(FM >= 4, "need >=1 m-rep per phase");
  constexpr int HA = BM / 2;          // A half-tile rows
  constexpr int HB = BN / 2;          // B half-tile rows
  constexpr int LB = HB / 64;         // loads/lane per B half-tile
  __shared__ bf16_t As[2][BM * 64];
  __shared__ bf16_t Bs[2][BN * 64];

  // Bijective XCD-aware swizzle (8 XCDs).
  const int nwg = mtiles * ntiles;
  const int orig = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7, xcd = orig & 7;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) +
                 (orig >> 3);
  const int tile_m = wg / ntiles, tile_n = wg % ntiles;
  const int row0 = tile_m * BM, col0 = tile_n * BN;

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wm = wid >> 2, wn = wid & 3;  // 2 x 4 wave grid

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int NT = K / 64;  // K-tiles (host guarantees K % 64 == 0)

  // ---- prologue: tile 0 fully + B(1); leave B(1) pair in flight ----
  stage_half<HA>(A, lda, row0, M, 0, &As[0][0], wid, lane);
  stage_half<HA>(A, lda, row0 + HA, M, 0, &As[0][HA * 64], wid, lane);
  stage_half<HB>(B, ldb, col0, N, 0, &Bs[0][0], wid, lane);
  stage_half<HB>(B, ldb, col0 + HB, N, 0, &Bs[0][HB * 64], wid, lane);
  if (NT > 1) {
    stage_half<HB>(B, ldb, col0, N, 64, &Bs[1][0], wid, lane);
    stage_half<HB>(B, ldb, col0 + HB, N, 64, &Bs[1][HB * 64], wid, lane);
    if (SAFE) vmcnt_wait<0>(); else vmcnt_wait<2 * LB>();
  } else {
    vmcnt_wait<0>();
  }
  S_BARRIER();

  const int arow_base = wm * (FM * 16) + (lane & 15);
  const int brow_base = wn * (FN * 16) + (lane & 15);
  const int kblk = (lane >> 4);  // 8-element k-block within a 32-k-step

  for (int t = 0; t < NT; ++t) {
    const int buf = t & 1;
    const bf16_t* at = &As[buf][0];
    const bf16_t* bt = &Bs[buf][0];
    s16x8 bfrag[FN][2];
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      // --- ds-load this phase's register subtile ---
      s16x8 afrag[MG][2];
#pragma unroll
      for (int g = 0; g < MG; ++g) {
        const int m = p * MG + g;
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          afrag[g][kk] =
              lds_frag(at, arow_base + m * 16, kk * 4 + kblk);
      }
      if (p == 0) {
#pragma unroll
        for (int n = 0; n < FN; ++n)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            bfrag[n][kk] =
                lds_frag(bt, brow_base + n * 16, kk * 4 + kblk);
      }
      // --- issue this phase's half-tile prefetch ---
      if (p == 0 && t + 1 < NT) {
        stage_half<HA>(A, lda, row0, M, (t + 1) * 64, &As[buf ^ 1][0], wid,
                       lane);
      } else if (p == 1 && t + 1 < NT) {
        stage_half<HA>(A, lda, row0 + HA, M, (t + 1) * 64,
                       &As[buf ^ 1][HA * 64], wid, lane);
      } else if (p == 2 && t + 2 < NT) {
        stage_half<HB>(B, ldb, col0, N, (t + 2) * 64, &Bs[buf][0], wid,
                       lane);
      } else if (p == 3 && t + 2 < NT) {
        stage_half<HB>(B, ldb, col0 + HB, N, (t + 2) * 64,
                       &Bs[buf][HB * 64], wid, lane);
      }
      if (p == 3) {
        // Once per K-tile: everything except the B(t+2) pair must land.
        if (SAFE) vmcnt_wait<0>(); else vmcnt_wait<2 * LB>();
      }
      if (SAFE) {
        __syncthreads();
      } else {
        S_BARRIER();
      }
      // asm ds_read results land only after lgkmcnt; the sched_barrier
      // keeps the compiler from hoisting the (register-only) MFMAs above
      // the wait (guide rule 18).
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      // --- MFMA cluster ---
      if (SETPRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int g = 0; g < MG; ++g) {
        const int m = p * MG + g;
#pragma unroll
        for (int n = 0; n < FN; ++n)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[g][kk], bfrag[n][kk], acc[m][n], 0, 0, 0);
      }
      if (SETPRIO) __builtin_amdgcn_s_setprio(0);
      if (SAFE) {
        __syncthreads();
      } else {
        S_BARRIER();
      }
    }
  }

  // ---- epilogue: C/D layout col = lane&15, row = (lane>>4)*4 + reg ----
  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < FM; ++i) {
#pragma unroll
    for (int j = 0; j < FN; ++j) {
      const int col = col0 + wn * (FN * 16) + j * 16 + c_col_in_frag;
      if (col >= N) continue;
      const float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int row = row0 + wm * (FM * 16) + i * 16 + c_row_base + rr;
        if (row >= M) continue;
        float v = acc[i][j][rr] + bv;
        if (act == 1) v = v > 0.f ? v : 0.f;
        if (act == 3) {  // fused relu+dropout (stateless counter RNG)
          v = v > 0.f ? v : 0.f;
          const uint32_t r =
              hash_rng((uint64_t)seed[0], (uint64_t)row * N + col);
          v = (r >= pthresh) ? v * inv_keep : 0.f;
        }
        bf16_t* cp = &C[(int64_t)row * ldc + col];
        if (act == 2) v += bf2f(*cp);
        *cp = f2bf(v);
      }
    }
  }
}

// s_waitcnt lgkmcnt(N) with a compile-time literal.
template <int N>
__device__ __forceinline__ void lgkm_wait() {
  static_assert(N == 0 || N == 2 || N == 4, "unsupported lgkmcnt");
  if constexpr (N == 0) {
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  } else if constexpr (N == 2) {
    asm volatile("s_waitcnt lgkmcnt(2)" ::: "memory");
  } else if constexpr (N == 4) {
    asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory");
  }
}

// Register-pipelined variant: fragment ds_reads are issued one phase AHEAD
// of their MFMA (counted lgkmcnt instead of a full drain), one barrier per
// phase instead of two, and every read targets the CURRENT K-tile:
//   ph0: issue B(all) + A(m-groups 0,1); lgkm leaves group 1 in flight
//   ph1: issue A(group 2);               lgkm leaves group 2
//   ph2: issue A(group 3);               lgkm leaves group 3
//   ph3: no reads; lgkm(0); counted vmcnt BEFORE the barrier so the whole
//        next tile is landed when its ph0 reads issue.
// Prefetch is unchanged (A(t+1) at ph0/1, B(t+2) at ph2/3).
template <int BM, int BN, bool SETPRIO = false>
__global__ __launch_bounds__(512, 2) void gemm_nt8p_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, int64_t lda, int64_t ldb, int64_t ldc, int act, int mtiles,
    int ntiles) {
  constexpr int FM = BM / (2 * 16);
  constexpr int FN = BN / (4 * 16);
  constexpr int MG = FM / 4;  // m-reps per phase
  static_assert(FM >= 4, "need >=1 m-rep per phase");
  constexpr int HA = BM / 2;
  constexpr int HB = BN / 2;
  constexpr int LB = HB / 64;  // loads/lane per B half-tile
  __shared__ bf16_t As[2][BM * 64];
  __shared__ bf16_t Bs[2][BN * 64];

  const int nwg = mtiles * ntiles;
  const int orig = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7, xcd = orig & 7;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) +
                 (orig >> 3);
  const int tile_m = wg / ntiles, tile_n = wg % ntiles;
  const int row0 = tile_m * BM, col0 = tile_n * BN;

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wm = wid >> 2, wn = wid & 3;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int NT = K / 64;

  stage_half<HA>(A, lda, row0, M, 0, &As[0][0], wid, lane);
  stage_half<HA>(A, lda, row0 + HA, M, 0, &As[0][HA * 64], wid, lane);
  stage_half<HB>(B, ldb, col0, N, 0, &Bs[0][0], wid, lane);
  stage_half<HB>(B, ldb, col0 + HB, N, 0, &Bs[0][HB * 64], wid, lane);
  if (NT > 1) {
    stage_half<HB>(B, ldb, col0, N, 64, &Bs[1][0], wid, lane);
    stage_half<HB>(B, ldb, col0 + HB, N, 64, &Bs[1][HB * 64], wid, lane);
    vmcnt_wait<2 * LB>();
  } else {
    vmcnt_wait<0>();
  }
  S_BARRIER();

  const int arow_base = wm * (FM * 16) + (lane & 15);
  const int brow_base = wn * (FN * 16) + (lane & 15);
  const int kblk = (lane >> 4);

  s16x8 bfrag[FN][2];
  s16x8 afrag[2][MG][2];  // [phase parity][m-rep in group][k-step]

  for (int t = 0; t < NT; ++t) {
    const int buf = t & 1;
    const bf16_t* at = &As[buf][0];
    const bf16_t* bt = &Bs[buf][0];
    const bool pre_b0 = t + 2 < NT;

    // ---- ph0: issue B(all) + A groups 0,1; compute group 0 ----
#pragma unroll
    for (int n = 0; n < FN; ++n)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        bfrag[n][kk] = lds_frag(bt, brow_base + n * 16, kk * 4 + kblk);
#pragma unroll
    for (int g = 0; g < MG; ++g)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        afrag[0][g][kk] =
            lds_frag(at, arow_base + g * 16, kk * 4 + kblk);
#pragma unroll
    for (int g = 0; g < MG; ++g)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        afrag[1][g][kk] =
            lds_frag(at, arow_base + (MG + g) * 16, kk * 4 + kblk);
    if (t + 1 < NT)
      stage_half<HA>(A, lda, row0, M, (t + 1) * 64, &As[buf ^ 1][0], wid,
                     lane);
    lgkm_wait<2 * MG>();  // group 1 still in flight
    __builtin_amdgcn_sched_barrier(0);
    if (SETPRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int g = 0; g < MG; ++g)
#pragma unroll
      for (int n = 0; n < FN; ++n)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[g][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[0][g][kk], bfrag[n][kk], acc[g][n], 0, 0, 0);
    if (SETPRIO) __builtin_amdgcn_s_setprio(0);
    S_BARRIER();

    // ---- ph1/ph2: issue group p+1; compute group p ----
#pragma unroll
    for (int p = 1; p < 3; ++p) {
      const int cur = p & 1, nxt = cur ^ 1;
#pragma unroll
      for (int g = 0; g < MG; ++g)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          afrag[nxt][g][kk] = lds_frag(
              at, arow_base + ((p + 1) * MG + g) * 16, kk * 4 + kblk);
      if (p == 1) {
        if (t + 1 < NT)
          stage_half<HA>(A, lda, row0 + HA, M, (t + 1) * 64,
                         &As[buf ^ 1][HA * 64], wid, lane);
      } else {
        if (pre_b0)
          stage_half<HB>(B, ldb, col0, N, (t + 2) * 64, &Bs[buf][0], wid,
                         lane);
      }
      lgkm_wait<2 * MG>();
      __builtin_amdgcn_sched_barrier(0);
      if (SETPRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int g = 0; g < MG; ++g)
#pragma unroll
        for (int n = 0; n < FN; ++n)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            acc[p * MG + g][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[cur][g][kk], bfrag[n][kk], acc[p * MG + g][n], 0, 0,
                0);
      if (SETPRIO) __builtin_amdgcn_s_setprio(0);
      S_BARRIER();
    }

    // ---- ph3: no reads; land the whole next tile; compute group 3 ----
    if (pre_b0) {
      vmcnt_wait<LB>();  // leave only B0(t+2) in flight
    } else {
      vmcnt_wait<0>();
    }
    if (t + 2 < NT)
      stage_half<HB>(B, ldb, col0 + HB, N, (t + 2) * 64, &Bs[buf][HB * 64],
                     wid, lane);
    lgkm_wait<0>();
    __builtin_amdgcn_sched_barrier(0);
    if (SETPRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int g = 0; g < MG; ++g)
#pragma unroll
      for (int n = 0; n < FN; ++n)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[3 * MG + g][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[1][g][kk], bfrag[n][kk], acc[3 * MG + g][n], 0, 0, 0);
    if (SETPRIO) __builtin_amdgcn_s_setprio(0);
    S_BARRIER();
  }

  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < FM; ++i) {
#pragma unroll
    for (int j = 0; j < FN; ++j) {
      const int col = col0 + wn * (FN * 16) + j * 16 + c_col_in_frag;
      if (col >= N) continue;
      const float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int row = row0 + wm * (FM * 16) + i * 16 + c_row_base + rr;
        if (row >= M) continue;
        float v = acc[i][j][rr] + bv;
        if (act == 1) v = v > 0.f ? v : 0.f;
        bf16_t* cp = &C[(int64_t)row * ldc + col];
        if (act == 2) v += bf2f(*cp);
        *cp = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------
// K-slab staged kernel: the deepest pipeline variant. BK=64 is staged as
// TWO [rows x 32] column slabs per operand; phases {0,1} compute k-slab 0,
// {2,3} k-slab 1, so consumption is K-progressive and each staged unit
// has 5-6 phases to land (the half-tile schedule above gives A(t+1) only
// 2 phases — its vmcnt stalls when loaded-latency HBM exceeds that):
//
//   issue at tile t:  ph0: A-slab1(t+1)   ph1: B-slab1(t+1)
//                     ph2: A-slab0(t+2)   ph3: B-slab0(t+2)
//   (slab-s LDS homes are free one whole tile early: slab0 reads end at
//    ph1, so slab0(t+2) can be issued at t ph2.)
//   waits: vmcnt(4 units) at ph1-end (guards slab1(t) reads at ph2) and
//   ph3-end (guards slab0(t+1) reads at t+1 ph0) — 4 units stay in
//   flight across EVERY barrier.
//
// Slab rows are 64 B; fragment ds_read_b128 at (row, 16B-block q=lane>>4)
// would be 8-way bank-conflicted; the involution blk ^= (row>>1)&3
// spreads the wave across all 32 banks at 2-way (free). Stage-side the
// same XOR is applied to the per-lane GLOBAL source block
// ((lane&3) ^ ((lane>>3)&3) — linear LDS dest as global_load_lds needs).
// ---------------------------------------------------------------------

// Stage one [ROWS x 32] K-slab (rows of 64 B; NW*16 rows per block round).
template <int ROWS, int NW>
__device__ __forceinline__ void stage_slab(
    const bf16_t* __restrict__ G, int64_t ld, int rows0, int max_row, int k0,
    bf16_t* __restrict__ lds_slab, int wid, int lane) {
#pragma unroll
  for (int r = 0; r < ROWS / (NW * 16); ++r) {
    const int row_in = r * (NW * 16) + wid * 16 + (lane >> 2);
    int grow = rows0 + row_in;
    grow = grow < max_row ? grow : max_row - 1;
    const int src_blk = (lane & 3) ^ ((lane >> 3) & 3);
    const bf16_t* gp = G + (int64_t)grow * ld + k0 + src_blk * 8;
    bf16_t* lp = lds_slab + (r * (NW * 16) + wid * 16) * 32;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)gp,
        (__attribute__((address_space(3))) void*)lp, 16, 0, 0);
  }
}

// K-major slab staging: [COLS/16][32][16] subtiles, one 1 KiB DMA each
// (the gemm_tn.hip tr scheme per 32-k slab).
typedef __attribute__((ext_vector_type(2))) unsigned int u32x2_k;

__device__ __forceinline__ u32x2_k ks_tr_read(const bf16_t* p) {
  const unsigned off = (unsigned)(uintptr_t)(
      const __attribute__((address_space(3))) bf16_t*)p;
  u32x2_k out;
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(out) : "v"(off));
  return out;
}

template <int COLS, int NW>
__device__ __forceinline__ void stage_slab_tr(
    const bf16_t* __restrict__ G, int64_t ld, int col0, int max_col, int k0,
    int max_k, bf16_t* __restrict__ lds_slab, int wid, int lane) {
  constexpr int SUBT = COLS / 16;
#pragma unroll
  for (int st = wid; st < SUBT; st += NW) {
    const int k = k0 + (lane >> 1);
    int col = col0 + st * 16 + (lane & 1) * 8;
    if (col + 8 > max_col) col = max(0, (max_col - 8) & ~7);
    const bf16_t* gp = G + (int64_t)min(k, max_k - 1) * ld + col;
    bf16_t* lp = lds_slab + st * 512;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)gp,
        (__attribute__((address_space(3))) void*)lp, 16, 0, 0);
  }
}

// Fragment from a K-major slab: two hardware-transpose reads of the
// [32][16] subtile (measured cooperative semantics, gemm_tn.hip).
__device__ __forceinline__ s16x8 lds_frag_ks_tr(const bf16_t* lds_slab,
                                                int colblock, int lane) {
  const bf16_t* sub = lds_slab + colblock * 512;
  const int row = 8 * (lane >> 4) + ((lane >> 2) & 3);
  const int col4 = 4 * (lane & 3);
  u32x2_k r0 = ks_tr_read(sub + row * 16 + col4);
  u32x2_k r1 = ks_tr_read(sub + (row + 4) * 16 + col4);
  union {
    s16x8 f;
    struct { u32x2_k lo, hi; } u;
  } pack;
  pack.u.lo = r0;
  pack.u.hi = r1;
  return pack.f;
}

__device__ __forceinline__ s16x8 lds_frag_ks(const bf16_t* lds_slab,
                                             int row, int q) {
  const int blk = q ^ ((row >> 1) & 3);
  const uint32_t addr = (uint32_t)(uintptr_t)(
      const __attribute__((address_space(3))) void*)
      &lds_slab[row * 32 + blk * 8];
  s16x8 out;
  asm volatile("ds_read_b128 %0, %1" : "=v"(out) : "v"(addr));
  return out;
}

// BAR2=false drops the barrier between a slab's two compute phases:
// both phases read the SAME already-published slab, and the next sync
// point (vmcnt + barrier at the slab-pair end) re-aligns the waves —
// phase-internal desync has no hazard (reads race only reads; stage
// targets are disjoint from any in-flight reads).
template <int BM, int BN, int WN, bool TRA = false, bool TRB = false,
          bool BAR2 = false,  // single barrier per slab measured +2-16%
          bool STAGGER = false>  // persistent even-wave priority boost
__global__ __launch_bounds__(2 * WN * 64, 2) void gemm_ks_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, int64_t lda, int64_t ldb, int64_t ldc, int act, int mtiles,
    int ntiles, const long long* __restrict__ seed = nullptr,
    uint32_t pthresh = 0, float inv_keep = 1.f) {
  constexpr int NW = 2 * WN;
  constexpr int FM = BM / 32;          // m-fragments per wave (2 M-waves)
  constexpr int FN = BN / (WN * 16);
  constexpr int MH = FM / 2;           // m-reps per phase
  static_assert(FM >= 2, "need >=1 m-rep per phase pair");
  constexpr int LPU_A = BM / (NW * 16);  // loads/lane per A slab
  constexpr int LPU_B = BN / (NW * 16);
  constexpr int INFLIGHT = 2 * LPU_A + 2 * LPU_B;  // 4 units
  __shared__ bf16_t As[2][2][BM * 32];  // [dbuf][slab]
  __shared__ bf16_t Bs[2][2][BN * 32];

  const int nwg = mtiles * ntiles;
  const int orig = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7, xcd = orig & 7;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) +
                 (orig >> 3);
  const int tile_m = wg / ntiles, tile_n = wg % ntiles;
  const int row0 = tile_m * BM, col0 = tile_n * BN;

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wm = wid / WN, wn = wid % WN;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // Wave-stagger experiment: the barriers keep SIMD-sharing waves in
  // lockstep, so both hit ds_read phases together and the MFMA pipe
  // idles; a persistent priority split lets the scheduler drift them.
  if (STAGGER && (wid & 1) == 0) __builtin_amdgcn_s_setprio(1);

  const int NT = K / 64;

#define KS_STAGE_A(buf, slab, t_)                                             \
  do {                                                                        \
    if (TRA)                                                                  \
      stage_slab_tr<BM, NW>(A, lda, row0, M, (t_)*64 + (slab)*32, K,          \
                            &As[buf][slab][0], wid, lane);                    \
    else                                                                      \
      stage_slab<BM, NW>(A, lda, row0, M, (t_)*64 + (slab)*32,                \
                         &As[buf][slab][0], wid, lane);                       \
  } while (0)
#define KS_STAGE_B(buf, slab, t_)                                             \
  do {                                                                        \
    if (TRB)                                                                  \
      stage_slab_tr<BN, NW>(B, ldb, col0, N, (t_)*64 + (slab)*32, K,          \
                            &Bs[buf][slab][0], wid, lane);                    \
    else                                                                      \
      stage_slab<BN, NW>(B, ldb, col0, N, (t_)*64 + (slab)*32,                \
                         &Bs[buf][slab][0], wid, lane);                       \
  } while (0)

  // prologue: tile0 both slabs + tile1 slab0; leave 4 units in flight.
  KS_STAGE_A(0, 0, 0);
  KS_STAGE_B(0, 0, 0);
  KS_STAGE_A(0, 1, 0);
  KS_STAGE_B(0, 1, 0);
  if (NT > 1) {
    KS_STAGE_A(1, 0, 1);
    KS_STAGE_B(1, 0, 1);
    vmcnt_wait<INFLIGHT>();
  } else {
    vmcnt_wait<0>();
  }
  S_BARRIER();

  const int arow_base = wm * (FM * 16) + (lane & 15);
  const int brow_base = wn * (FN * 16) + (lane & 15);
  const int kq = lane >> 4;

  for (int t = 0; t < NT; ++t) {
    const int buf = t & 1;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const bf16_t* at = &As[buf][kk][0];
      const bf16_t* bt = &Bs[buf][kk][0];
      s16x8 bfrag[FN];
#pragma unroll
      for (int ph = 0; ph < 2; ++ph) {  // global phase p = kk*2 + ph
        s16x8 afrag[MH];
        if (ph == 0) {
#pragma unroll
          for (int n = 0; n < FN; ++n)
            bfrag[n] = TRB ? lds_frag_ks_tr(bt, wn * FN + n, lane)
                           : lds_frag_ks(bt, brow_base + n * 16, kq);
        }
#pragma unroll
        for (int g = 0; g < MH; ++g)
          afrag[g] = TRA
              ? lds_frag_ks_tr(at, wm * FM + ph * MH + g, lane)
              : lds_frag_ks(at, arow_base + (ph * MH + g) * 16, kq);
        // prefetch issue: p0: A-s1(t+1), p1: B-s1(t+1),
        //                 p2: A-s0(t+2), p3: B-s0(t+2).
        // ALWAYS issued, with the tile index clamped at the tail: skipping
        // would shrink the outstanding-load count and turn the fixed
        // vmcnt(INFLIGHT) waits vacuous exactly when the next tile's slab0
        // is still in flight (caught by the 4096 race screen on the LPU=1
        // config). Clamped dummy loads land in slots no later tile reads.
        if (kk == 0) {
          const int tn = t + 1 < NT ? t + 1 : NT - 1;
          if (ph == 0) {
            KS_STAGE_A(buf ^ 1, 1, tn);
          } else {
            KS_STAGE_B(buf ^ 1, 1, tn);
          }
        } else {
          const int tn = t + 2 < NT ? t + 2 : NT - 1;
          if (ph == 0) {
            KS_STAGE_A(buf, 0, tn);
          } else {
            KS_STAGE_B(buf, 0, tn);
          }
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int g = 0; g < MH; ++g)
#pragma unroll
          for (int n = 0; n < FN; ++n)
            acc[ph * MH + g][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[g], bfrag[n], acc[ph * MH + g][n], 0, 0, 0);
        if (ph == 1) {
          // end of the k-slab pair: publish the next consumers' data
          // (kk0: slab1(t) reads at p2; kk1: slab0(t+1) reads at t+1 p0)
          vmcnt_wait<INFLIGHT>();
          S_BARRIER();
        } else if (BAR2) {
          S_BARRIER();
        }
      }
    }
  }
#undef KS_STAGE_A
#undef KS_STAGE_B

  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < FM; ++i) {
#pragma unroll
    for (int j = 0; j < FN; ++j) {
      const int col = col0 + wn * (FN * 16) + j * 16 + c_col_in_frag;
      if (col >= N) continue;
      const float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int row = row0 + wm * (FM * 16) + i * 16 + c_row_base + rr;
        if (row >= M) continue;
        float v = acc[i][j][rr] + bv;
        if (act == 1) v = v > 0.f ? v : 0.f;
        if (act == 3) {  // fused relu+dropout (stateless counter RNG)
          v = v > 0.f ? v : 0.f;
          const uint32_t rnd =
              hash_rng((uint64_t)seed[0], (uint64_t)row * N + col);
          v = (rnd >= pthresh) ? v * inv_keep : 0.f;
        }
        bf16_t* cp = &C[(int64_t)row * ldc + col];
        if (act == 2) v += bf2f(*cp);
        *cp = f2bf(v);
      }
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------
// Generalized pipelined kernel: same 4-phase schedule as gemm_nt8p_kernel,
// additionally templated over the wave grid (2 x WN) and per-operand
// K-major staging (TRA/TRB) using the ds_read_b64_tr_b16 hardware
// transpose (semantics measured in gemm_tn.hip round 1). This brings the
// counted-vmcnt pipeline to the backward dX (trans_b) / dW (tt) GEMMs,
// which the round-2 bench profile shows at ~40% of step GPU time on the
// drain-bound 2-phase structure (profiles/prof_r02b).
// ---------------------------------------------------------------------

typedef __attribute__((ext_vector_type(2))) unsigned int u32x2_g;

__device__ __forceinline__ u32x2_g tr8_b16_read(const bf16_t* p) {
  const unsigned off = (unsigned)(uintptr_t)(
      const __attribute__((address_space(3))) bf16_t*)p;
  u32x2_g out;
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(out) : "v"(off));
  return out;
}

namespace {

// Linear stage of one half-tile generalized over wave count.
template <int HROWS, int NW>
__device__ __forceinline__ void stage_half_g(
    const bf16_t* __restrict__ G, int64_t ld, int rows0, int max_row, int k0,
    bf16_t* __restrict__ lds_half, int wid, int lane) {
#pragma unroll
  for (int r = 0; r < HROWS / (NW * 8); ++r) {
    const int row_in_half = r * (NW * 8) + wid * 8 + (lane >> 3);
    int grow = rows0 + row_in_half;
    grow = grow < max_row ? grow : max_row - 1;
    const int src_blk = (lane & 7) ^ ((lane >> 3) & 7);
    const bf16_t* gp = G + (int64_t)grow * ld + k0 + src_blk * 8;
    bf16_t* lp = lds_half + (r * (NW * 8) + wid * 8) * 64;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)gp,
        (__attribute__((address_space(3))) void*)lp, 16, 0, 0);
  }
}

// K-major stage of one half-tile (HCOLS output-cols x 64 k) from G[k][col]:
// layout [2 kslab][HCOLS/16][32][16], one 1 KiB DMA per [32][16] subtile.
template <int HCOLS, int NW>
__device__ __forceinline__ void stage_half_tr8(
    const bf16_t* __restrict__ G, int64_t ld, int col0, int max_col, int k0,
    int max_k, bf16_t* __restrict__ lds_half, int wid, int lane) {
  constexpr int SUBT = HCOLS / 16;
#pragma unroll
  for (int slab = 0; slab < 2; ++slab) {
#pragma unroll
    for (int st = wid; st < SUBT; st += NW) {
      const int k = k0 + slab * 32 + (lane >> 1);
      int col = col0 + st * 16 + (lane & 1) * 8;
      if (col + 8 > max_col) col = max(0, (max_col - 8) & ~7);
      const bf16_t* gp = G + (int64_t)min(k, max_k - 1) * ld + col;
      bf16_t* lp = lds_half + slab * HCOLS * 32 + st * 512;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)gp,
          (__attribute__((address_space(3))) void*)lp, 16, 0, 0);
    }
  }
}

// Fragment from a K-major-staged half: two hardware-transpose reads of the
// [32][16] subtile (measured cooperative-transpose semantics, gemm_tn.hip).
template <int HCOLS>
__device__ __forceinline__ s16x8 frag_tr8(const bf16_t* lds_half,
                                          int colblock, int kk, int lane) {
  const bf16_t* sub = lds_half + kk * HCOLS * 32 + colblock * 512;
  const int row = 8 * (lane >> 4) + ((lane >> 2) & 3);
  const int col4 = 4 * (lane & 3);
  u32x2_g r0 = tr8_b16_read(sub + row * 16 + col4);
  u32x2_g r1 = tr8_b16_read(sub + (row + 4) * 16 + col4);
  union {
    s16x8 f;
    struct { u32x2_g lo, hi; } u;
  } pack;
  pack.u.lo = r0;
  pack.u.hi = r1;
  return pack.f;
}

template <int N>
__device__ __forceinline__ void lgkm_wait_g() {
  static_assert(N == 0 || N == 2 || N == 4 || N == 8, "lgkm count");
  if constexpr (N == 0) asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  else if constexpr (N == 2) asm volatile("s_waitcnt lgkmcnt(2)" ::: "memory");
  else if constexpr (N == 4) asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory");
  else if constexpr (N == 8) asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
}

// Pipelined 4-phase kernel, generalized: 2 x WN wave grid, per-operand
// K-major staging. Schedule identical to gemm_nt8p_kernel (see above).
template <int BM, int BN, int WN, bool TRA, bool TRB>
__global__ __launch_bounds__(2 * WN * 64, 2) void gemm_x8_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, int64_t lda, int64_t ldb, int64_t ldc, int act, int mtiles,
    int ntiles) {
  constexpr int NW = 2 * WN;
  constexpr int FM = BM / (2 * 16);
  constexpr int FN = BN / (WN * 16);
  constexpr int MG = FM / 4;
  static_assert(FM >= 4, "need >=1 m-rep per phase");
  constexpr int HA = BM / 2;
  constexpr int HB = BN / 2;
  constexpr int LB = HB / (8 * NW) > 0 ? HB / (8 * NW) : 1;  // loads/lane/half
  __shared__ bf16_t As[2][BM * 64];
  __shared__ bf16_t Bs[2][BN * 64];

  const int nwg = mtiles * ntiles;
  const int orig = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7, xcd = orig & 7;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) +
                 (orig >> 3);
  const int tile_m = wg / ntiles, tile_n = wg % ntiles;
  const int row0 = tile_m * BM, col0 = tile_n * BN;

  const int tid = threadIdx.x;
  const int wid = tid >> 6, lane = tid & 63;
  const int wm = wid / WN, wn = wid % WN;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int NT = K / 64;

#define STAGE_A8(buf, half, k0)                                               \
  do {                                                                        \
    if (TRA)                                                                  \
      stage_half_tr8<HA, NW>(A, lda, row0 + (half)*HA, M, k0, K,              \
                             &As[buf][(half)*HA * 64], wid, lane);            \
    else                                                                      \
      stage_half_g<HA, NW>(A, lda, row0 + (half)*HA, M, k0,                   \
                           &As[buf][(half)*HA * 64], wid, lane);              \
  } while (0)
#define STAGE_B8(buf, half, k0)                                               \
  do {                                                                        \
    if (TRB)                                                                  \
      stage_half_tr8<HB, NW>(B, ldb, col0 + (half)*HB, N, k0, K,              \
                             &Bs[buf][(half)*HB * 64], wid, lane);            \
    else                                                                      \
      stage_half_g<HB, NW>(B, ldb, col0 + (half)*HB, N, k0,                   \
                           &Bs[buf][(half)*HB * 64], wid, lane);              \
  } while (0)

  STAGE_A8(0, 0, 0);
  STAGE_A8(0, 1, 0);
  STAGE_B8(0, 0, 0);
  STAGE_B8(0, 1, 0);
  if (NT > 1) {
    STAGE_B8(1, 0, 64);
    STAGE_B8(1, 1, 64);
    vmcnt_wait<2 * LB>();
  } else {
    vmcnt_wait<0>();
  }
  S_BARRIER();

  const int arow_base = wm * (FM * 16) + (lane & 15);
  const int brow_base = wn * (FN * 16) + (lane & 15);
  const int kblk = (lane >> 4);

  // Fragment loaders: operand half selected by the wave's fixed span.
  // A: wave wm's rows live in half wm (WARPS_M == 2). B: half = cols
  // span / HB. For K-major operands the fragment col-block indexes into
  // the wave's half-local subtile array.
#define LOAD_A8(dst, at_, m)                                                  \
  do {                                                                        \
    if (TRA) {                                                                \
      const int cb = (wm * FM + (m)) % (HA / 16);                             \
      const bf16_t* hbase = (at_) + ((wm * FM + (m)) / (HA / 16)) * HA * 64;  \
      (dst)[0] = frag_tr8<HA>(hbase, cb, 0, lane);                            \
      (dst)[1] = frag_tr8<HA>(hbase, cb, 1, lane);                            \
    } else {                                                                  \
      (dst)[0] = lds_frag(at_, arow_base + (m)*16, kblk);                     \
      (dst)[1] = lds_frag(at_, arow_base + (m)*16, 4 + kblk);                 \
    }                                                                         \
  } while (0)
#define LOAD_B8(dst, bt_, n)                                                  \
  do {                                                                        \
    if (TRB) {                                                                \
      const int cb = (wn * FN + (n)) % (HB / 16);                             \
      const bf16_t* hbase = (bt_) + ((wn * FN + (n)) / (HB / 16)) * HB * 64;  \
      (dst)[0] = frag_tr8<HB>(hbase, cb, 0, lane);                            \
      (dst)[1] = frag_tr8<HB>(hbase, cb, 1, lane);                            \
    } else {                                                                  \
      (dst)[0] = lds_frag(bt_, brow_base + (n)*16, kblk);                     \
      (dst)[1] = lds_frag(bt_, brow_base + (n)*16, 4 + kblk);                 \
    }                                                                         \
  } while (0)

  s16x8 bfrag[FN][2];
  s16x8 afrag[2][MG][2];
  constexpr int AG_READS = MG * 2 * (TRA ? 2 : 1);  // lgkm units per A group

  for (int t = 0; t < NT; ++t) {
    const int buf = t & 1;
    const bf16_t* at = &As[buf][0];
    const bf16_t* bt = &Bs[buf][0];
    const bool pre_b0 = t + 2 < NT;

    // ph0
#pragma unroll
    for (int n = 0; n < FN; ++n) LOAD_B8(bfrag[n], bt, n);
#pragma unroll
    for (int g = 0; g < MG; ++g) LOAD_A8(afrag[0][g], at, g);
#pragma unroll
    for (int g = 0; g < MG; ++g) LOAD_A8(afrag[1][g], at, MG + g);
    if (t + 1 < NT) STAGE_A8(buf ^ 1, 0, (t + 1) * 64);
    lgkm_wait_g<AG_READS>();
    __builtin_amdgcn_sched_barrier(0);
#pragma unroll
    for (int g = 0; g < MG; ++g)
#pragma unroll
      for (int n = 0; n < FN; ++n)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[g][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[0][g][kk], bfrag[n][kk], acc[g][n], 0, 0, 0);
    S_BARRIER();

    // ph1, ph2
#pragma unroll
    for (int p = 1; p < 3; ++p) {
      const int cur = p & 1, nxt = cur ^ 1;
#pragma unroll
      for (int g = 0; g < MG; ++g)
        LOAD_A8(afrag[nxt][g], at, (p + 1) * MG + g);
      if (p == 1) {
        if (t + 1 < NT) STAGE_A8(buf ^ 1, 1, (t + 1) * 64);
      } else {
        if (pre_b0) STAGE_B8(buf, 0, (t + 2) * 64);
      }
      lgkm_wait_g<AG_READS>();
      __builtin_amdgcn_sched_barrier(0);
#pragma unroll
      for (int g = 0; g < MG; ++g)
#pragma unroll
        for (int n = 0; n < FN; ++n)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            acc[p * MG + g][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[cur][g][kk], bfrag[n][kk], acc[p * MG + g][n], 0, 0,
                0);
      S_BARRIER();
    }

    // ph3
    if (pre_b0) {
      vmcnt_wait<LB>();
    } else {
      vmcnt_wait<0>();
    }
    if (t + 2 < NT) STAGE_B8(buf, 1, (t + 2) * 64);
    lgkm_wait_g<0>();
    __builtin_amdgcn_sched_barrier(0);
#pragma unroll
    for (int g = 0; g < MG; ++g)
#pragma unroll
      for (int n = 0; n < FN; ++n)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[3 * MG + g][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[1][g][kk], bfrag[n][kk], acc[3 * MG + g][n], 0, 0, 0);
    S_BARRIER();
  }
#undef STAGE_A8
#undef STAGE_B8
#undef LOAD_A8
#undef LOAD_B8

  const int c_col_in_frag = lane & 15;
  const int c_row_base = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < FM; ++i) {
#pragma unroll
    for (int j = 0; j < FN; ++j) {
      const int col = col0 + wn * (FN * 16) + j * 16 + c_col_in_frag;
      if (col >= N) continue;
      const float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int row = row0 + wm * (FM * 16) + i * 16 + c_row_base + rr;
        if (row >= M) continue;
        float v = acc[i][j][rr] + bv;
        if (act == 1) v = v > 0.f ? v : 0.f;
        bf16_t* cp = &C[(int64_t)row * ldc + col];
        if (act == 2) v += bf2f(*cp);
        *cp = f2bf(v);
      }
    }
  }
}

}  // namespace

// Probe/production entry for the generalized pipelined kernel.
// A is [M,K] ([K,M] when trans_a); B is [N,K] ([K,N] when trans_b).
// variant: 0 = 128^2 8-wave, 1 = 128^2 4-wave, 2 = 256^2 8-wave,
//          3 = 256x128 8-wave, 4 = 128x256 8-wave.
void gemm_x8(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
             const c10::optional<at::Tensor>& bias, int64_t act,
             int64_t trans_a, int64_t trans_b, int64_t variant) {
  const int M = (int)C.size(0), N = (int)C.size(1);
  const int K = (int)(trans_a ? A.size(0) : A.size(1));
  TORCH_CHECK((int)(trans_b ? B.size(1) : B.size(0)) == N &&
              (int)(trans_b ? B.size(0) : B.size(1)) == K &&
              (int)(trans_a ? A.size(1) : A.size(0)) == M,
              "gemm_x8: shape mismatch");
  TORCH_CHECK(K % 64 == 0, "gemm_x8: K % 64");
  const int64_t lda = A.stride(0), ldb = B.stride(0), ldc = C.stride(0);
  TORCH_CHECK(lda % 8 == 0 && ldb % 8 == 0, "gemm_x8: 16B-aligned rows");
  const float* bias_ptr = nullptr;
  if (bias.has_value() && bias->defined()) bias_ptr = bias->data_ptr<float>();
  auto stream = at::cuda::getCurrentCUDAStream();
  const bf16_t* a = (const bf16_t*)A.data_ptr();
  const bf16_t* b = (const bf16_t*)B.data_ptr();
  bf16_t* c = (bf16_t*)C.data_ptr();

#define LX8(BM, BN, WNW, TA, TB)                                              \
  do {                                                                        \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    hipLaunchKernelGGL((gemm_x8_kernel<BM, BN, WNW, TA, TB>), dim3(mt * nt),  \
                       dim3(2 * WNW * 64), 0, stream.stream(), a, b, c,       \
                       bias_ptr, M, N, K, lda, ldb, ldc, (int)act, mt, nt);   \
  } while (0)
#define LX8_T(BM, BN, WNW)                                                    \
  do {                                                                        \
    if (trans_a && trans_b) LX8(BM, BN, WNW, true, true);                     \
    else if (trans_b) LX8(BM, BN, WNW, false, true);                          \
    else if (trans_a) LX8(BM, BN, WNW, true, false);                          \
    else LX8(BM, BN, WNW, false, false);                                      \
  } while (0)
  switch (variant) {
    case 0: LX8_T(128, 128, 4); break;
    case 1: LX8_T(128, 128, 2); break;
    case 2: LX8_T(256, 256, 4); break;
    case 3: LX8_T(256, 128, 4); break;
    case 4: LX8_T(128, 256, 4); break;
#define LKS_T(BM, BN, WNW)                                                    \
  do {                                                                        \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    if (trans_a && trans_b) {                                                 \
      hipLaunchKernelGGL((gemm_ks_kernel<BM, BN, WNW, true, true>),           \
                         dim3(mt * nt), dim3(2 * WNW * 64), 0,                \
                         stream.stream(), a, b, c, bias_ptr, M, N, K, lda,    \
                         ldb, ldc, (int)act, mt, nt);                         \
    } else if (trans_b) {                                                     \
      hipLaunchKernelGGL((gemm_ks_kernel<BM, BN, WNW, false, true>),          \
                         dim3(mt * nt), dim3(2 * WNW * 64), 0,                \
                         stream.stream(), a, b, c, bias_ptr, M, N, K, lda,    \
                         ldb, ldc, (int)act, mt, nt);                         \
    } else if (trans_a) {                                                     \
      hipLaunchKernelGGL((gemm_ks_kernel<BM, BN, WNW, true, false>),          \
                         dim3(mt * nt), dim3(2 * WNW * 64), 0,                \
                         stream.stream(), a, b, c, bias_ptr, M, N, K, lda,    \
                         ldb, ldc, (int)act, mt, nt);                         \
    } else {                                                                  \
      hipLaunchKernelGGL((gemm_ks_kernel<BM, BN, WNW, false, false>),         \
                         dim3(mt * nt), dim3(2 * WNW * 64), 0,                \
                         stream.stream(), a, b, c, bias_ptr, M, N, K, lda,    \
                         ldb, ldc, (int)act, mt, nt);                         \
    }                                                                         \
  } while (0)
    case 10: LKS_T(128, 128, 4); break;
    case 11: LKS_T(256, 256, 4); break;
    case 12: LKS_T(256, 128, 4); break;
#undef LKS_T
    default: TORCH_CHECK(false, "gemm_x8: unknown variant");
  }
#undef LX8
#undef LX8_T
  HIP_CHECK_KERNEL();
}

// Production dispatch hook, called from gemm.hip's gemm_nt_bf16 fast path.
// Returns false when the 8-phase kernels don't apply (caller falls back).
// Thresholds from measured A/B (profiles/gemm8_r02c.json, MI355X):
//   * 256^2 noprio wins at big grids (1036-1167 TF vs old 860-914 @4k/8k),
//   * 128^2 8-phase wins at 2048-class grids (552 vs 473 TF @2048^3),
//   * setprio measured NEGATIVE on this lockstep schedule (-17%): off.
bool gemm_nt8_try(const bf16_t* a, const bf16_t* b, bf16_t* c,
                  const float* bias_ptr, int M, int N, int K, int64_t lda,
                  int64_t ldb, int64_t ldc, int act, hipStream_t stream) {
  if (K % 64 != 0 || K < 128 || lda % 8 || ldb % 8) return false;
  const int64_t t256 = (int64_t)((M + 255) / 256) * ((N + 255) / 256);
  const int64_t t128 = (int64_t)((M + 127) / 128) * ((N + 127) / 128);
  if (t256 >= 200) {
    const int mt = (M + 255) / 256, nt = (N + 255) / 256;
    hipLaunchKernelGGL((gemm_nt8_kernel<256, 256, false, false>),
                       dim3(mt * nt), dim3(512), 0, stream, a, b, c,
                       bias_ptr, M, N, K, lda, ldb, ldc, act, mt, nt);
    return true;
  }
  if (t128 >= 256) {
    // K-slab schedule: 643-687 TF at the 2048-class bench shapes vs 555
    // for the half-tile 8-phase (profiles/gemm_ks_r02b.json, same-box).
    const int mt = (M + 127) / 128, nt = (N + 127) / 128;
    hipLaunchKernelGGL((gemm_ks_kernel<128, 128, 4>), dim3(mt * nt),
                       dim3(512), 0, stream, a, b, c, bias_ptr, M, N, K,
                       lda, ldb, ldc, act, mt, nt);
    return true;
  }
  return false;
}

// Dropout twin of gemm_nt8_try: same tile thresholds, act==3 epilogue.
bool gemm_nt8_try_dropout(const bf16_t* a, const bf16_t* b, bf16_t* c,
                          const float* bias_ptr, int M, int N, int K,
                          int64_t lda, int64_t ldb, int64_t ldc,
                          const long long* seed, uint32_t pthresh,
                          float inv_keep, hipStream_t stream) {
  if (K % 64 != 0 || K < 128 || lda % 8 || ldb % 8) return false;
  const int64_t t256 = (int64_t)((M + 255) / 256) * ((N + 255) / 256);
  const int64_t t128 = (int64_t)((M + 127) / 128) * ((N + 127) / 128);
  if (t256 >= 200) {
    const int mt = (M + 255) / 256, nt = (N + 255) / 256;
    hipLaunchKernelGGL((gemm_nt8_kernel<256, 256, false, false>),
                       dim3(mt * nt), dim3(512), 0, stream, a, b, c,
                       bias_ptr, M, N, K, lda, ldb, ldc, 3, mt, nt, seed,
                       pthresh, inv_keep);
    return true;
  }
  if (t128 >= 256) {
    const int mt = (M + 127) / 128, nt = (N + 127) / 128;
    hipLaunchKernelGGL((gemm_ks_kernel<128, 128, 4>), dim3(mt * nt),
                       dim3(512), 0, stream, a, b, c, bias_ptr, M, N, K,
                       lda, ldb, ldc, 3, mt, nt, seed, pthresh, inv_keep);
    return true;
  }
  return false;
}

// Host entry: launches the 8-phase kernel for a given tile config.
// variant: 0 = 256x256, 1 = 256x128, 2 = 128x256, 3 = 128x128,
//          +10 = SAFE (drain) twin, +20 = no-setprio twin.
void gemm_nt8(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
              const c10::optional<at::Tensor>& bias, int64_t act,
              int64_t variant) {
  TORCH_CHECK(A.is_cuda() && B.is_cuda() && C.is_cuda(), "gemm8: GPU only");
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16, "gemm8: bf16 inputs");
  const int M = (int)A.size(0), K = (int)A.size(1), N = (int)B.size(0);
  TORCH_CHECK(B.size(1) == K && C.size(0) == M && C.size(1) == N,
              "gemm8: shape mismatch");
  TORCH_CHECK(K % 64 == 0, "gemm8: K % 64 != 0");
  const int64_t lda = A.stride(0), ldb = B.stride(0), ldc = C.stride(0);
  TORCH_CHECK(A.stride(1) == 1 && B.stride(1) == 1 && C.stride(1) == 1 &&
              lda % 8 == 0 && ldb % 8 == 0, "gemm8: need 16B-aligned rows");
  const float* bias_ptr = nullptr;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->scalar_type() == at::kFloat && bias->numel() == N,
                "gemm8: bias must be fp32[N]");
    bias_ptr = bias->data_ptr<float>();
  }
  auto stream = at::cuda::getCurrentCUDAStream();
  const bf16_t* a = (const bf16_t*)A.data_ptr();
  const bf16_t* b = (const bf16_t*)B.data_ptr();
  bf16_t* c = (bf16_t*)C.data_ptr();

#define LAUNCH_8PH(BM, BN, SAFE, PRIO)                                        \
  do {                                                                        \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    hipLaunchKernelGGL((gemm_nt8_kernel<BM, BN, SAFE, PRIO>), dim3(mt * nt),  \
                       dim3(512), 0, stream.stream(), a, b, c, bias_ptr, M,   \
                       N, K, lda, ldb, ldc, (int)act, mt, nt);                \
  } while (0)

  switch (variant) {
    case 0: LAUNCH_8PH(256, 256, false, true); break;
    case 1: LAUNCH_8PH(256, 128, false, true); break;
    case 2: LAUNCH_8PH(128, 256, false, true); break;
    case 3: LAUNCH_8PH(128, 128, false, true); break;
    case 10: LAUNCH_8PH(256, 256, true, true); break;
    case 11: LAUNCH_8PH(256, 128, true, true); break;
    case 12: LAUNCH_8PH(128, 256, true, true); break;
    case 13: LAUNCH_8PH(128, 128, true, true); break;
    case 20: LAUNCH_8PH(256, 256, false, false); break;
    case 23: LAUNCH_8PH(128, 128, false, false); break;
#define LAUNCH_8PHP(BM, BN, PRIO)                                             \
  do {                                                                        \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    hipLaunchKernelGGL((gemm_nt8p_kernel<BM, BN, PRIO>), dim3(mt * nt),       \
                       dim3(512), 0, stream.stream(), a, b, c, bias_ptr, M,   \
                       N, K, lda, ldb, ldc, (int)act, mt, nt);                \
  } while (0)
#define LAUNCH_KS(BM, BN, WNW)                                               \
  do {                                                                        \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    hipLaunchKernelGGL((gemm_ks_kernel<BM, BN, WNW>), dim3(mt * nt),          \
                       dim3(2 * WNW * 64), 0, stream.stream(), a, b, c,       \
                       bias_ptr, M, N, K, lda, ldb, ldc, (int)act, mt, nt);   \
  } while (0)
    case 50: LAUNCH_KS(256, 256, 4); break;
    case 51: LAUNCH_KS(128, 128, 4); break;
    case 52: LAUNCH_KS(128, 128, 2); break;
    case 53: LAUNCH_KS(256, 128, 4); break;
    case 55: {
      const int mt = (M + 255) / 256, nt = (N + 255) / 256;
      hipLaunchKernelGGL(
          (gemm_ks_kernel<256, 256, 4, false, false, false>), dim3(mt * nt),
          dim3(512), 0, stream.stream(), a, b, c, bias_ptr, M, N, K, lda,
          ldb, ldc, (int)act, mt, nt);
      break;
    }
    case 56: {
      const int mt = (M + 127) / 128, nt = (N + 127) / 128;
      hipLaunchKernelGGL(
          (gemm_ks_kernel<128, 128, 4, false, false, false>), dim3(mt * nt),
          dim3(512), 0, stream.stream(), a, b, c, bias_ptr, M, N, K, lda,
          ldb, ldc, (int)act, mt, nt);
      break;
    }
    case 57: {  // 256^2 + even-wave priority stagger
      const int mt = (M + 255) / 256, nt = (N + 255) / 256;
      hipLaunchKernelGGL(
          (gemm_ks_kernel<256, 256, 4, false, false, false, true>),
          dim3(mt * nt), dim3(512), 0, stream.stream(), a, b, c, bias_ptr,
          M, N, K, lda, ldb, ldc, (int)act, mt, nt);
      break;
    }
    case 58: {  // 128^2 8w + stagger
      const int mt = (M + 127) / 128, nt = (N + 127) / 128;
      hipLaunchKernelGGL(
          (gemm_ks_kernel<128, 128, 4, false, false, false, true>),
          dim3(mt * nt), dim3(512), 0, stream.stream(), a, b, c, bias_ptr,
          M, N, K, lda, ldb, ldc, (int)act, mt, nt);
      break;
    }
#undef LAUNCH_KS
    case 30: LAUNCH_8PHP(256, 256, false); break;
    case 31: LAUNCH_8PHP(256, 256, true); break;
    case 32: LAUNCH_8PHP(256, 128, false); break;
    case 33: LAUNCH_8PHP(128, 256, false); break;
    case 34: LAUNCH_8PHP(128, 128, false); break;
#undef LAUNCH_8PHP
    default: TORCH_CHECK(false, "gemm8: unknown variant");
  }
#undef LAUNCH_8PH
  HIP_CHECK_KERNEL();
}
