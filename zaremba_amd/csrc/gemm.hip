// MFMA bf16 GEMM kernels for gfx950 (CDNA4).
//
// Covers the framework's dense-GEMM call sites:
//   K2  input-side gate GEMM   gx = x @ W_x^T        (NT, reference model.py:35)
//   K6  output projection      scores = h @ fc.W^T   (NT, reference model.py:67)
//   K8  backward data GEMMs    dx = dG @ W (NT via pre-transposed shadow
//       weights) and weight-grad GEMMs dW = dG^T @ x (TN).
//
// Design notes (measured on MI355X, see profiles/):
//   * BMxBN output tile (128x128 or 64x64 chosen by grid size so skinny
//     shapes still fill 256 CUs), BK=64, 4 waves in a 2x2 grid,
//     v_mfma_f32_16x16x32_bf16 with fp32 accumulation.
//   * Register-staged LDS with the T14 async-split schedule: issue tile
//     t+1's global loads immediately after publishing tile t to LDS, so
//     HBM/L2 latency hides under the MFMA phase.
//   * Interior blocks take a guard-free staging path — per-element bounds
//     branches around global loads de-pipeline hipcc's vmcnt bookkeeping
//     (one drained load per element), measured 3-5x slower.
//   * LDS images are [row][k] with byte ^= (row&7)<<4 XOR swizzle:
//     conflict-reduced ds_read_b128 column-slice reads.
//   * The TN (transposed) staging loads 8 k-rows per thread with
//     lane-coalesced strided reads and writes one b128 per thread.
#include "common.h"

namespace zamd {

constexpr int BK = 64;
constexpr int GEMM_THREADS = 256;

template <int BROWS>
DEV_INLINE int swz(int row, int byte_col) {
  return row * (BK * 2) + (byte_col ^ ((row & 7) << 4));
}

// ---- staging: k-contiguous operand (NT layout), [BROWS][BK] tile ---------
// Register-load phase: each thread grabs BROWS*BK/(256*8) bf16x8 vectors.
template <int BROWS, bool GUARD>
DEV_INLINE void stage_load_kc(const bf16* __restrict__ src, int ld, int row0,
                              int nrows, int k0, int K,
                              bf16x8 (&v)[BROWS / 32]) {
  const int t = threadIdx.x;
  const int kk = (t & 7) * 8;
  const int r = t >> 3;
#pragma unroll
  for (int p = 0; p < BROWS / 32; ++p) {
    const int row = r + p * 32;
    if (GUARD) {
      v[p] = bf16x8{};
      const int gr = row0 + row;
      const int gk = k0 + kk;
      if (row < nrows) {
        const bf16* sp = src + (int64_t)gr * ld + gk;
        if (gk + 8 <= K) {
          v[p] = *reinterpret_cast<const bf16x8*>(sp);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            v[p][e] = (gk + e < K) ? sp[e] : (bf16)0.f;
        }
      }
    } else {
      v[p] = *reinterpret_cast<const bf16x8*>(src + (int64_t)(row0 + row) * ld +
                                              k0 + kk);
    }
  }
}

template <int BROWS>
DEV_INLINE void stage_write_kc(bf16* lds, bf16x8 (&v)[BROWS / 32]) {
  const int t = threadIdx.x;
  const int kk = (t & 7) * 8;
  const int r = t >> 3;
#pragma unroll
  for (int p = 0; p < BROWS / 32; ++p)
    *reinterpret_cast<bf16x8*>(reinterpret_cast<char*>(lds) +
                               swz<BROWS>(r + p * 32, kk * 2)) = v[p];
}

// ---- staging: transposed operand (TN layout: src[K, rows]) ----------------
template <int BROWS, bool GUARD>
DEV_INLINE void stage_load_tr(const bf16* __restrict__ src, int ld, int row0,
                              int nrows, int k0, int K,
                              bf16x8 (&v)[BROWS / 32]) {
  const int t = threadIdx.x;
  const int row = t & (BROWS - 1);
  const int kb = (t / BROWS) * 8;           // 256/BROWS k-groups per pass
  constexpr int KG = 8 * (256 / BROWS);     // k covered per pass
  const int gr = row0 + row;
#pragma unroll
  for (int p = 0; p < BROWS / 32; ++p) {    // BK/KG passes == BROWS/32
    const int kk = kb + p * KG;
    if (GUARD) {
      v[p] = bf16x8{};
      if (row < nrows) {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int gk = k0 + kk + e;
          if (gk < K) v[p][e] = src[(int64_t)gk * ld + gr];
        }
      }
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e)
        v[p][e] = src[(int64_t)(k0 + kk + e) * ld + gr];
    }
  }
}

template <int BROWS>
DEV_INLINE void stage_write_tr(bf16* lds, bf16x8 (&v)[BROWS / 32]) {
  const int t = threadIdx.x;
  const int row = t & (BROWS - 1);
  const int kb = (t / BROWS) * 8;
  constexpr int KG = 8 * (256 / BROWS);
#pragma unroll
  for (int p = 0; p < BROWS / 32; ++p)
    *reinterpret_cast<bf16x8*>(reinterpret_cast<char*>(lds) +
                               swz<BROWS>(row, (kb + p * KG) * 2)) = v[p];
}

template <int BROWS>
DEV_INLINE bf16x8 frag_read(const bf16* lds, int row, int k) {
  return *reinterpret_cast<const bf16x8*>(
      reinterpret_cast<const char*>(lds) + swz<BROWS>(row, k * 2));
}

// ---------------------------------------------------------------------------
// C[M,N] = A' @ B' + bias. TRANS_A: A is [K,M]; TRANS_B: B is [K,N];
// otherwise the k-contiguous layouts A[M,K] / B[N,K]. BM=BN=TILE; 4 waves,
// each (TILE/2)x(TILE/2).
// ---------------------------------------------------------------------------
template <int TILE, bool TRANS_A, bool TRANS_B, typename OutT>
__global__ __launch_bounds__(GEMM_THREADS) void gemm_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    OutT* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, int lda, int ldb, int ldc) {
  constexpr int WT = TILE / 2;       // wave tile (64 or 32)
  constexpr int NF = WT / 16;        // fragments per wave dim (4 or 2)
  __shared__ bf16 As[TILE * BK];
  __shared__ bf16 Bs[TILE * BK];

  const int nbn = (N + TILE - 1) / TILE;
  const int bm = blockIdx.x / nbn;
  const int bn = blockIdx.x % nbn;
  const int m0 = bm * TILE, n0 = bn * TILE;
  const bool interior =
      (m0 + TILE <= M) && (n0 + TILE <= N);

  const int w = wave_id();
  const int wm = (w >> 1) * WT;
  const int wn = (w & 1) * WT;
  const int l = lane_id();
  const int lm = l & 15;
  const int lk = (l >> 4) * 8;

  f32x4 acc[NF][NF] = {};
  bf16x8 va[TILE / 32], vb[TILE / 32];

  const int nk = (K + BK - 1) / BK;
  const int k_full = K / BK;         // tiles with no K guard

  auto load_tile = [&](int kt, bool guard_mn) {
    const bool gk = (kt >= k_full);
    if (!guard_mn && !gk) {
      if (TRANS_A)
        stage_load_tr<TILE, false>(A, lda, m0, TILE, kt * BK, K, va);
      else
        stage_load_kc<TILE, false>(A, lda, m0, TILE, kt * BK, K, va);
      if (TRANS_B)
        stage_load_tr<TILE, false>(B, ldb, n0, TILE, kt * BK, K, vb);
      else
        stage_load_kc<TILE, false>(B, ldb, n0, TILE, kt * BK, K, vb);
    } else {
      if (TRANS_A)
        stage_load_tr<TILE, true>(A, lda, m0, min(TILE, M - m0), kt * BK, K, va);
      else
        stage_load_kc<TILE, true>(A, lda, m0, min(TILE, M - m0), kt * BK, K, va);
      if (TRANS_B)
        stage_load_tr<TILE, true>(B, ldb, n0, min(TILE, N - n0), kt * BK, K, vb);
      else
        stage_load_kc<TILE, true>(B, ldb, n0, min(TILE, N - n0), kt * BK, K, vb);
    }
  };

  // The k-contiguous path software-pipelines (T14: hold tile kt+1 in
  // registers across the MFMA phase). The transposed path stages
  // synchronously — its 64 scalar-loaded values per thread would push the
  // kernel to 278 registers (1 wave/SIMD) if held across the MFMA phase.
  constexpr bool ASYNC = !(TRANS_A || TRANS_B);
  if (ASYNC) load_tile(0, !interior);
  for (int kt = 0; kt < nk; ++kt) {
    if (!ASYNC) load_tile(kt, !interior);
    __syncthreads();  // LDS consumers of tile kt-1 done
    if (TRANS_A)
      stage_write_tr<TILE>(As, va);
    else
      stage_write_kc<TILE>(As, va);
    if (TRANS_B)
      stage_write_tr<TILE>(Bs, vb);
    else
      stage_write_kc<TILE>(Bs, vb);
    if (ASYNC && kt + 1 < nk) load_tile(kt + 1, !interior);  // issue early
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      bf16x8 af[NF], bfr[NF];
#pragma unroll
      for (int i = 0; i < NF; ++i)
        af[i] = frag_read<TILE>(As, wm + i * 16 + lm, kk + lk);
#pragma unroll
      for (int j = 0; j < NF; ++j)
        bfr[j] = frag_read<TILE>(Bs, wn + j * 16 + lm, kk + lk);
#pragma unroll
      for (int i = 0; i < NF; ++i)
#pragma unroll
        for (int j = 0; j < NF; ++j)
          acc[i][j] = mfma_16x16x32_bf16(af[i], bfr[j], acc[i][j]);
    }
  }

  // Epilogue. C/D map: col = l&15, row = (l>>4)*4 + r.
  const int fr0 = (l >> 4) * 4;
#pragma unroll
  for (int j = 0; j < NF; ++j) {
    const int col = n0 + wn + j * 16 + (l & 15);
    if (col >= N) continue;
    const float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int i = 0; i < NF; ++i) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm + i * 16 + fr0 + r;
        if (row < M) C[(int64_t)row * ldc + col] = (OutT)(acc[i][j][r] + bv);
      }
    }
  }
}

template <bool TA, bool TB, typename OutT>
void launch_gemm_t(const bf16* A, const bf16* B, OutT* C, const float* bias,
                   int M, int N, int K, int lda, int ldb, int ldc,
                   hipStream_t stream) {
  // pick the tile: prefer 128^2; fall back to 64^2 when the grid would
  // under-fill the 256 CUs (skinny backward shapes)
  int grid128 = cdiv(M, 128) * cdiv(N, 128);
  if (grid128 >= 192) {
    hipLaunchKernelGGL((gemm_kernel<128, TA, TB, OutT>), dim3(grid128),
                       dim3(GEMM_THREADS), 0, stream, A, B, C, bias, M, N, K,
                       lda, ldb, ldc);
  } else {
    int grid64 = cdiv(M, 64) * cdiv(N, 64);
    hipLaunchKernelGGL((gemm_kernel<64, TA, TB, OutT>), dim3(grid64),
                       dim3(GEMM_THREADS), 0, stream, A, B, C, bias, M, N, K,
                       lda, ldb, ldc);
  }
}

#define INST(TA, TB, T)                                                     \
  template void launch_gemm_t<TA, TB, T>(const bf16*, const bf16*, T*,      \
                                         const float*, int, int, int, int,  \
                                         int, int, hipStream_t);
INST(false, false, float)
INST(false, false, bf16)
INST(true, true, float)
INST(true, true, bf16)
#undef INST

}  // namespace zamd
