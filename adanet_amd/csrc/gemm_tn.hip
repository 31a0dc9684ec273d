// Transposed-operand GEMM staging via ds_read_b64_tr_b16 (gfx950 hardware
// transpose read) — lets the backward GEMMs consume K-major operands
// (dz/W/x as stored) with NO separate transpose kernels:
//
//   dX[B,Kin]  = dz[B,Nout] @ W[Nout,Kin]   -> B-operand transposed
//   dW[N,Kin]  = dz^T @ x                   -> BOTH operands transposed
//
// Scheme per transposed operand (KSTEP=32):
//   * LDS holds [n_colblocks][32 k][16 cols] subtiles. One global_load_lds
//     (64 lanes x 16 B = 1 KiB) stages EXACTLY one subtile: lane l fetches
//     G[k0 + l/2][col0 + blk*16 + (l%2)*8 .. +8] (per-lane source, linear
//     wave-uniform LDS dest) — same op count as normal staging.
//   * fragment read: mfma b-frag lane l needs T[k=(l>>4)*8+j][l&15],
//     j=0..7 — built by two ds_read_b64_tr_b16 ops using the MEASURED
//     cooperative-transpose semantics (see tr_frag_addrs; verified by
//     probe_tr16_layout + random-data refchecks).
//
// Inline-asm ds_read needs an explicit lgkmcnt(0) + sched_barrier(0) fence
// before the MFMAs (the compiler does not order register-only MFMAs
// against asm loads — CDNA4 guide rule 18).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

typedef s16x8 frag_ab;
typedef __attribute__((ext_vector_type(2))) unsigned int u32x2;

// Pipelined counted-vmcnt path (gemm_8ph.hip) — measured winner for the
// trans_b-only (dX) mid shapes: +9-20% over the 2-phase structure at
// 2048-class grids (profiles/gemm_x8_r02a.json).
void gemm_x8(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
             const c10::optional<at::Tensor>& bias, int64_t act,
             int64_t trans_a, int64_t trans_b, int64_t variant);

typedef __attribute__((address_space(3))) const bf16_t* lds_cptr_t;

__device__ __forceinline__ u32x2 tr_b16_read(const bf16_t* lds_half_addr) {
  // AS(3) pointers are 32-bit LDS offsets on AMDGPU; the addrspacecast
  // recovers the DS-instruction operand from the generic pointer.
  const unsigned off = (unsigned)(uintptr_t)(lds_cptr_t)lds_half_addr;
  u32x2 out;
  asm volatile("ds_read_b64_tr_b16 %0, %1"
               : "=v"(out)
               : "v"(off)
               : "memory");
  return out;
}

// MEASURED (tr_probe rounds 1-2): ds_read_b64_tr_b16 is a 16-lane-group
// cooperative transpose. Each lane mu loads 4 contiguous halfwords at its
// own address; the result redistributes them as
//     result(l, j) = data(lane 16*(l>>4) + 4*j + ((l>>2)&3)) [l&3].
// For an mfma fragment from T[32 k][16 c] (lane l needs T[8g+j][l&15],
// j=0..7) the source-lane address must therefore be
//     A(mu) = (8*(mu>>4) + ((mu>>2)&3) + 4*phase)*16 + 4*(mu&3)
// with phase 0/1 for j=0..3 / 4..7 (A2 = A1 + 64 halfwords).
__device__ __forceinline__ void tr_frag_addrs(const bf16_t* sub, int lane,
                                              const bf16_t** a1,
                                              const bf16_t** a2) {
  const int row = 8 * (lane >> 4) + ((lane >> 2) & 3);
  const int col4 = 4 * (lane & 3);
  *a1 = sub + row * 16 + col4;
  *a2 = sub + (row + 4) * 16 + col4;
}

// ---------------------------------------------------------------- probe
// Writes the tr-read of a ramp-filled [32][16] subtile so python can verify
// the lane->element mapping before trusting the GEMM.
__global__ void probe_tr16_layout_kernel(float* __restrict__ out) {
  __shared__ bf16_t tile[32 * 16];
  const int lane = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 32 * 16; i += blockDim.x) {
    tile[i] = f2bf((float)(i % 256));  // bf16-exact ramp
  }
  __syncthreads();
  const bf16_t *a1, *a2;
  tr_frag_addrs(tile, lane, &a1, &a2);
  u32x2 r0 = tr_b16_read(a1);
  u32x2 r1 = tr_b16_read(a2);
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
  const unsigned short* h0 = (const unsigned short*)&r0;
  const unsigned short* h1 = (const unsigned short*)&r1;
  if (blockIdx.x == 0) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const unsigned short a = h0[j];
      const unsigned short b = h1[j];
      out[lane * 8 + j] = bits2f(*(const short*)&a);
      out[lane * 8 + 4 + j] = bits2f(*(const short*)&b);
    }
  }
}

void probe_tr16_layout(at::Tensor& out) {
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(probe_tr16_layout_kernel, dim3(1), dim3(64), 0,
                     stream.stream(), out.data_ptr<float>());
  HIP_CHECK_KERNEL();
}

// ------------------------------------------------------- transposed stage
// Stage a [ROWS x 32] K-minor logical tile whose GLOBAL source is K-major
// (G[k][col]): one 1 KiB op per [32][16] subtile, wave w takes subtiles
// strided by NWAVES.
template <int ROWS, int NWAVES>
__device__ __forceinline__ void stage_tile_tr(
    const bf16_t* __restrict__ G, int ld, int tile_col0, int max_col, int k0,
    int max_k, bf16_t* __restrict__ lds, int wid, int lane) {
  constexpr int SUBTILES = ROWS / 16;
#pragma unroll
  for (int st = wid; st < SUBTILES; st += NWAVES) {
    const int k = k0 + (lane >> 1);
    int col = tile_col0 + st * 16 + (lane & 1) * 8;
    if (col + 8 > max_col) {
      // clamp to the last 8-aligned window (masked on C-store anyway).
      col = max(0, (max_col - 8) & ~7);
    }
    const bf16_t* gp = G + (int64_t)min(k, max_k - 1) * ld + col;
    bf16_t* lp = lds + st * 32 * 16;  // wave-uniform subtile base
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)gp,
        (__attribute__((address_space(3))) void*)lp, 16, 0, 0);
  }
}

// Fragment from a transposed-staged operand: two quad-transpose reads.
__device__ __forceinline__ frag_ab frag_from_tr(const bf16_t* lds,
                                                int row_block, int lane) {
  const bf16_t* sub = lds + row_block * 32 * 16;  // 16 rows' subtile
  const bf16_t *a1, *a2;
  tr_frag_addrs(sub, lane, &a1, &a2);
  u32x2 r0 = tr_b16_read(a1);
  u32x2 r1 = tr_b16_read(a2);
  union {
    frag_ab f;
    struct { u32x2 lo, hi; } u;
  } pack;
  pack.u.lo = r0;
  pack.u.hi = r1;
  return pack.f;
}

// TRB: B-operand K-major (B'[K][N]); TRA additionally A K-major (A'[K][M]).
template <int BM, int BN, int FM, int FN, int MINWAVES, int WGM, int WGN,
          bool TRA, bool TRB, int GROUPM = 0>
__global__ __launch_bounds__(WGM * WGN * 64, MINWAVES) void gemm_tr_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, int lda, int ldb, int ldc, int act, int mtiles, int ntiles,
    int64_t stride_a = 0, int64_t stride_b = 0, int64_t stride_c = 0) {
  constexpr int KSTEP = 32;
  __shared__ bf16_t As[2][BM * KSTEP];
  __shared__ bf16_t Bs[2][BN * KSTEP];
  // Batched mode (pointwise conv: one GEMM per image): grid.z selects the
  // batch element; zero strides share the operand (the weight matrix).
  A += (int64_t)blockIdx.z * stride_a;
  B += (int64_t)blockIdx.z * stride_b;
  C += (int64_t)blockIdx.z * stride_c;
  const int nwg = mtiles * ntiles;
  const int orig = blockIdx.x;
  const int q = nwg >> 3, r = nwg & 7, xcd = orig & 7;
  const int wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) +
                 (orig >> 3);
  // GROUPM: L2 supertile grouping (see gemm.hip) — column-slab tile walk.
  int tile_m, tile_n;
  if (GROUPM > 0) {
    const int per_band = GROUPM * ntiles;
    const int band = wg / per_band;
    const int in_band = wg % per_band;
    const int gm = min(GROUPM, mtiles - band * GROUPM);
    tile_m = band * GROUPM + in_band % gm;
    tile_n = in_band / gm;
  } else {
    tile_m = wg / ntiles;
    tile_n = wg % ntiles;
  }
  static_assert(BM == WGM * FM * 16 && BN == WGN * FN * 16, "geometry");
  constexpr int NWAVES = WGM * WGN;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & 63;
  const int wm = wid / WGN, wn = wid % WGN;
  const int row0 = tile_m * BM;
  const int col0 = tile_n * BN;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int ktiles = K / KSTEP;

#define STAGE_A(buf, k0)                                                      \
  do {                                                                        \
    if (TRA)                                                                  \
      stage_tile_tr<BM, NWAVES>(A, lda, row0, M, k0, K, As[buf], wid, lane);  \
    else {                                                                    \
      constexpr int RPS = 512 / KSTEP;                                        \
      constexpr int LPR = KSTEP / 8;                                          \
      for (int seg = wid; seg < BM / RPS; seg += NWAVES) {                    \
        const int rt = seg * RPS + lane / LPR;                                \
        int grow = row0 + rt;                                                 \
        grow = grow < M ? grow : M - 1;                                       \
        const bf16_t* gp = A + (int64_t)grow * lda + k0 + (lane % LPR) * 8;   \
        bf16_t* lp = As[buf] + seg * RPS * KSTEP;                             \
        __builtin_amdgcn_global_load_lds(                                     \
            (const __attribute__((address_space(1))) void*)gp,                \
            (__attribute__((address_space(3))) void*)lp, 16, 0, 0);           \
      }                                                                       \
    }                                                                         \
  } while (0)
#define STAGE_B(buf, k0)                                                      \
  do {                                                                        \
    if (TRB)                                                                  \
      stage_tile_tr<BN, NWAVES>(B, ldb, col0, N, k0, K, Bs[buf], wid, lane);  \
    else {                                                                    \
      constexpr int RPS = 512 / KSTEP;                                        \
      constexpr int LPR = KSTEP / 8;                                          \
      for (int seg = wid; seg < BN / RPS; seg += NWAVES) {                    \
        const int rt = seg * RPS + lane / LPR;                                \
        int grow = col0 + rt;                                                 \
        grow = grow < N ? grow : N - 1;                                       \
        const bf16_t* gp = B + (int64_t)grow * ldb + k0 + (lane % LPR) * 8;   \
        bf16_t* lp = Bs[buf] + seg * RPS * KSTEP;                             \
        __builtin_amdgcn_global_load_lds(                                     \
            (const __attribute__((address_space(1))) void*)gp,                \
            (__attribute__((address_space(3))) void*)lp, 16, 0, 0);           \
      }                                                                       \
    }                                                                         \
  } while (0)

  STAGE_A(0, 0);
  STAGE_B(0, 0);
  int buf = 0;
  for (int kt = 0; kt < ktiles; ++kt) {
    // Explicit per-wave drain of the LDS-DMA issued last iteration:
    // __syncthreads() alone does not reliably order global_load_lds
    // completion against cross-wave LDS reads (the tr path reads via
    // inline-asm ds_read, invisible to the compiler's waitcnt insertion
    // -- observed as a run-to-run race in the NASNet 1x1 conv batched
    // GEMMs, benchmarks/nas_det_probe.py).
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (kt + 1 < ktiles) {
      STAGE_A(buf ^ 1, (kt + 1) * KSTEP);
      STAGE_B(buf ^ 1, (kt + 1) * KSTEP);
    }
    frag_ab a[FM], b[FN];
    const int kofs = (lane >> 4) * 8;
#pragma unroll
    for (int f = 0; f < FM; ++f) {
      if (TRA) {
        a[f] = frag_from_tr(As[buf], wm * FM + f, lane);
      } else {
        const int arow = wm * (FM * 16) + (lane & 15);
        a[f] = *(const frag_ab*)&As[buf][(arow + f * 16) * KSTEP + kofs];
      }
    }
#pragma unroll
    for (int f = 0; f < FN; ++f) {
      if (TRB) {
        b[f] = frag_from_tr(Bs[buf], wn * FN + f, lane);
      } else {
        const int brow = wn * (FN * 16) + (lane & 15);
        b[f] = *(const frag_ab*)&Bs[buf][(brow + f * 16) * KSTEP + kofs];
      }
    }
    if (TRA || TRB) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
    }
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
      for (int j = 0; j < FN; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], b[j],
                                                            acc[i][j], 0, 0,
                                                            0);
    buf ^= 1;
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
      // act==2: accumulate into C (direct-to-arena gradient writes — the
      // backward GEMM adds into the optimizer's grad view, replacing
      // autograd's separate AccumulateGrad add kernel).
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
#undef STAGE_A
#undef STAGE_B
}

// Probe entry: trans flags select operand layout. A is [M,K] (or [K,M] when
// trans_a), B is [N,K] (or [K,N] when trans_b); C[M,N].
void gemm_tr_probe(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
                   int64_t trans_a, int64_t trans_b, int64_t variant) {
  const int M = (int)C.size(0), N = (int)C.size(1);
  const int K = (int)(trans_a ? A.size(0) : A.size(1));
  const int lda = (int)A.stride(0), ldb = (int)B.stride(0),
            ldc = (int)C.stride(0);
  auto stream = at::cuda::getCurrentCUDAStream();
  const bf16_t* a = (const bf16_t*)A.data_ptr();
  const bf16_t* b = (const bf16_t*)B.data_ptr();
  bf16_t* c = (bf16_t*)C.data_ptr();
#define LV(BM, BN, FM, FN, MW, WGM, WGN, GM)                                  \
  do {                                                                        \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    if (trans_a && trans_b) {                                                 \
      hipLaunchKernelGGL(                                                     \
          (gemm_tr_kernel<BM, BN, FM, FN, MW, WGM, WGN, true, true, GM>),     \
          dim3(mt * nt), dim3(WGM * WGN * 64), 0, stream.stream(), a, b, c,  \
          nullptr, M, N, K, lda, ldb, ldc, 0, mt, nt);                        \
    } else if (trans_b) {                                                     \
      hipLaunchKernelGGL(                                                     \
          (gemm_tr_kernel<BM, BN, FM, FN, MW, WGM, WGN, false, true, GM>),    \
          dim3(mt * nt), dim3(WGM * WGN * 64), 0, stream.stream(), a, b, c,  \
          nullptr, M, N, K, lda, ldb, ldc, 0, mt, nt);                        \
    } else {                                                                  \
      hipLaunchKernelGGL(                                                     \
          (gemm_tr_kernel<BM, BN, FM, FN, MW, WGM, WGN, false, false, GM>),   \
          dim3(mt * nt), dim3(WGM * WGN * 64), 0, stream.stream(), a, b, c,  \
          nullptr, M, N, K, lda, ldb, ldc, 0, mt, nt);                        \
    }                                                                         \
  } while (0)
  switch (variant) {
    case 0: LV(128, 128, 2, 2, 4, 4, 4, 0); break;
    case 1: LV(128, 128, 2, 2, 4, 4, 4, 8); break;
    case 2: LV(64, 64, 2, 2, 6, 2, 2, 0); break;
    case 3: LV(64, 64, 2, 2, 6, 2, 2, 8); break;
    case 4: LV(256, 128, 4, 4, 2, 4, 2, 0); break;
    case 5: LV(128, 256, 4, 4, 2, 2, 4, 0); break;
    case 6: LV(128, 128, 2, 2, 4, 4, 4, 4); break;
    default: TORCH_CHECK(false, "unknown tr variant");
  }
#undef LV
  HIP_CHECK_KERNEL();
}

// Production entry for transposed-operand GEMMs (the backward dX/dW calls):
// C[M,N] = op(A) @ op(B)^T with A [K,M] when trans_a (else [M,K]) and
// B [K,N] when trans_b (else [N,K]). Geometry dispatch mirrors gemm.hip.
void gemm_tr_bf16(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
                  const c10::optional<at::Tensor>& bias, int64_t act,
                  int64_t trans_a, int64_t trans_b) {
  TORCH_CHECK(A.is_cuda() && B.is_cuda() && C.is_cuda(), "gemm_tr: GPU only");
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 &&
                  B.scalar_type() == at::kBFloat16,
              "gemm_tr: bf16 required");
  const int M = (int)C.size(0), N = (int)C.size(1);
  const int K = (int)(trans_a ? A.size(0) : A.size(1));
  TORCH_CHECK((int)(trans_b ? B.size(0) : B.size(1)) == K, "K mismatch");
  TORCH_CHECK((int)(trans_a ? A.size(1) : A.size(0)) == M, "M mismatch");
  TORCH_CHECK((int)(trans_b ? B.size(1) : B.size(0)) == N, "N mismatch");
  const int lda = (int)A.stride(0), ldb = (int)B.stride(0),
            ldc = (int)C.stride(0);
  TORCH_CHECK(trans_a || trans_b,
              "gemm_tr: use gemm_nt_bf16 for the non-transposed case");
  TORCH_CHECK(K % 32 == 0 && lda % 8 == 0 && ldb % 8 == 0,
              "gemm_tr: fast-path alignment required "
              "(K%32, strides%8) — pad like HipLinear does");
  if (M == 0 || N == 0) return;
  const float* bias_ptr = nullptr;
  if (bias.has_value() && bias->defined()) bias_ptr = bias->data_ptr<float>();
  auto stream = at::cuda::getCurrentCUDAStream();
  const bf16_t* a = (const bf16_t*)A.data_ptr();
  const bf16_t* b = (const bf16_t*)B.data_ptr();
  bf16_t* c = (bf16_t*)C.data_ptr();
  const int64_t b128 = (int64_t)((M + 127) / 128) * ((N + 127) / 128);
  const int64_t t_256x128 = (int64_t)((M + 255) / 256) * ((N + 127) / 128);
  const int64_t t_128x256 = (int64_t)((M + 127) / 128) * ((N + 255) / 256);
  // Pipelined/K-slab routing (same-box A/B, profiles/gemm_x8_r02b.json):
  //   dX (ft): K-slab wins at <=256-block grids (549 vs 441 @2048^2),
  //            the half-tile pipeline wins at larger grids (588 vs 515
  //            @2048x3072); measured at K=2048 AND 4096 (no K cap).
  //   dW (tt): K-slab 128^2 wins the 2048-class (528 vs 442, +19%);
  //            4096-class keeps the 2-phase 256x128 (986 vs 944).
  if (!trans_a && trans_b && K % 64 == 0 && b128 >= 256 && b128 < 1024) {
    gemm_x8(A, B, C, bias, act, trans_a, trans_b,
            b128 <= 256 ? /*K-slab 128^2*/ 10 : /*half-tile 128^2*/ 0);
    return;
  }
  if (trans_a && trans_b && K % 64 == 0 && b128 >= 256 && b128 < 1024) {
    gemm_x8(A, B, C, bias, act, trans_a, trans_b, /*K-slab 128^2*/ 10);
    return;
  }
  const int64_t t256sq = (int64_t)((M + 255) / 256) * ((N + 255) / 256);
  if (trans_a && trans_b && K % 64 == 0 && t256sq >= 200) {
    // 4096-class tt: K-slab 256^2 (bar1) 1033 vs 996 TF for the 2-phase
    // 256x128 (profiles/gemm_x8_r02c.json).
    gemm_x8(A, B, C, bias, act, trans_a, trans_b, /*K-slab 256^2*/ 11);
    return;
  }
#define LTR(BM, BN, FM, FN, MW, WGM, WGN)                                     \
  do {                                                                        \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    if (trans_a && trans_b) {                                                 \
      hipLaunchKernelGGL(                                                     \
          (gemm_tr_kernel<BM, BN, FM, FN, MW, WGM, WGN, true, true>),         \
          dim3(mt * nt), dim3(WGM * WGN * 64), 0, stream.stream(), a, b, c,  \
          bias_ptr, M, N, K, lda, ldb, ldc, (int)act, mt, nt);                \
    } else if (trans_b) {                                                     \
      hipLaunchKernelGGL(                                                     \
          (gemm_tr_kernel<BM, BN, FM, FN, MW, WGM, WGN, false, true>),        \
          dim3(mt * nt), dim3(WGM * WGN * 64), 0, stream.stream(), a, b, c,  \
          bias_ptr, M, N, K, lda, ldb, ldc, (int)act, mt, nt);                \
    } else {                                                                  \
      hipLaunchKernelGGL(                                                     \
          (gemm_tr_kernel<BM, BN, FM, FN, MW, WGM, WGN, true, false>),        \
          dim3(mt * nt), dim3(WGM * WGN * 64), 0, stream.stream(), a, b, c,  \
          bias_ptr, M, N, K, lda, ldb, ldc, (int)act, mt, nt);                \
    }                                                                         \
  } while (0)
  if (M <= 32 && N >= 256) {
    // skinny-M (classifier dW: M=n_classes=10, N=hidden): a 64-row tile
    // wastes >=half its MFMA work and fills M/64 of the row grid.
    if (N >= 96) LTR(32, 128, 2, 4, 4, 1, 2);
    else LTR(32, 64, 2, 2, 4, 1, 2);
  } else if (M >= N && t_256x128 >= 512) {
    // 1101 TF @4096^3 tt (profiles/tr_variants_r01.json)
    LTR(256, 128, 4, 4, 2, 4, 2);
  } else if (N > M && t_128x256 >= 512) {
    LTR(128, 256, 4, 4, 2, 2, 4);
  } else if (K >= M && K >= N && K <= 2048 && b128 >= 160 && b128 < 512) {
    // K-dominant MID shapes only, capped at the MEASURED territory
    // (K<=2048, grid under 512 tiles): 64^2 beats 128^2 by 14-22% at
    // 2048^3, both trans_b and tt (profiles/tr_variants_r01.json) — but
    // with K=4096 64^2 collapses (393-434 vs 620+ TF; the wide-hidden
    // bench regressed 11k->6.6k via dX 2048x3072x4096 before the K cap).
    LTR(64, 64, 2, 2, 6, 2, 2);
  } else if (b128 >= 160) {
    LTR(128, 128, 2, 2, 4, 4, 4);
  } else {
    LTR(64, 64, 2, 2, 6, 2, 2);
  }
#undef LTR
  HIP_CHECK_KERNEL();
}

// Batched transposed-operand GEMM: one GEMM per grid.z element (the
// pointwise-conv workhorse — K9: y_b[Co,HW] = W[Co,Ci] @ x_b[Ci,HW] is
// trans_b staging per image, dX_b = W^T @ dz_b is tt). 2-D operands are
// shared across the batch (stride 0); 3-D operands advance by stride(0).
void gemm_tr_batched(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
                     const c10::optional<at::Tensor>& bias, int64_t act,
                     int64_t trans_a, int64_t trans_b) {
  TORCH_CHECK(A.is_cuda() && B.is_cuda() && C.is_cuda(), "gemm_trb: GPU only");
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 &&
                  B.scalar_type() == at::kBFloat16,
              "gemm_trb: bf16 required");
  TORCH_CHECK(C.dim() == 3, "gemm_trb: C must be [Z,M,N]");
  const int Z = (int)C.size(0);
  const int M = (int)C.size(1), N = (int)C.size(2);
  auto dims2 = [](const at::Tensor& t) {
    return std::make_pair((int)t.size(t.dim() - 2), (int)t.size(t.dim() - 1));
  };
  auto a2 = dims2(A), b2 = dims2(B);
  const int K = trans_a ? a2.first : a2.second;
  TORCH_CHECK((trans_b ? b2.second : b2.first) == N, "gemm_trb: N mismatch");
  TORCH_CHECK((trans_b ? b2.first : b2.second) == K, "gemm_trb: K mismatch");
  TORCH_CHECK((trans_a ? a2.second : a2.first) == M, "gemm_trb: M mismatch");
  const int lda = (int)A.stride(A.dim() - 2), ldb = (int)B.stride(B.dim() - 2),
            ldc = (int)C.stride(1);
  const int64_t sa = A.dim() == 3 ? A.stride(0) : 0;
  const int64_t sb = B.dim() == 3 ? B.stride(0) : 0;
  const int64_t sc = C.stride(0);
  TORCH_CHECK(K % 32 == 0 && lda % 8 == 0 && ldb % 8 == 0,
              "gemm_trb: fast-path alignment required (K%32, strides%8)");
  if (Z == 0 || M == 0 || N == 0) return;
  const float* bias_ptr = nullptr;
  if (bias.has_value() && bias->defined()) bias_ptr = bias->data_ptr<float>();
  auto stream = at::cuda::getCurrentCUDAStream();
  const bf16_t* a = (const bf16_t*)A.data_ptr();
  const bf16_t* b = (const bf16_t*)B.data_ptr();
  bf16_t* c = (bf16_t*)C.data_ptr();
#define LTRB(BM, BN, FM, FN, MW, WGM, WGN, TRA, TRB)                          \
  do {                                                                        \
    const int mt = (M + BM - 1) / BM, nt = (N + BN - 1) / BN;                 \
    hipLaunchKernelGGL(                                                       \
        (gemm_tr_kernel<BM, BN, FM, FN, MW, WGM, WGN, TRA, TRB>),             \
        dim3(mt * nt, 1, Z), dim3(WGM * WGN * 64), 0, stream.stream(), a, b, \
        c, bias_ptr, M, N, K, lda, ldb, ldc, (int)act, mt, nt, sa, sb, sc);   \
  } while (0)
#define LTRB_DISPATCH(TRA, TRB)                                               \
  do {                                                                        \
    const int64_t b128 = (int64_t)((M + 127) / 128) * ((N + 127) / 128);      \
    if (M <= 32) {                                                            \
      /* skinny-M (pointwise conv with Co=32: a 64-row tile wastes half  */   \
      /* its MFMA work; 12.7% of the NASNet step ran on 64x64 tiles,     */   \
      /* profiles/nasprof7_summary.txt)                                  */   \
      if (N >= 96) LTRB(32, 128, 2, 4, 4, 1, 2, TRA, TRB);                    \
      else LTRB(32, 64, 2, 2, 4, 1, 2, TRA, TRB);                             \
    } else if (b128 >= 64) {                                                  \
      LTRB(128, 128, 2, 2, 4, 4, 4, TRA, TRB);                                \
    } else {                                                                  \
      LTRB(64, 64, 2, 2, 6, 2, 2, TRA, TRB);                                  \
    }                                                                         \
  } while (0)
  if (trans_a && trans_b) {
    LTRB_DISPATCH(true, true);
  } else if (trans_b) {
    LTRB_DISPATCH(false, true);
  } else if (trans_a) {
    LTRB_DISPATCH(true, false);
  } else {
    LTRB_DISPATCH(false, false);
  }
#undef LTRB
#undef LTRB_DISPATCH
  HIP_CHECK_KERNEL();
}
