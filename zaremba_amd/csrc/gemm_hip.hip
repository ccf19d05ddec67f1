#include "hip/hip_runtime.h"
// MFMA bf16 GEMM kernels for gfx950 (CDNA4).
//
// Covers the framework's dense-GEMM call sites:
//   K2  input-side gate GEMM   gx = x @ W_x^T        (NT, reference model.py:35)
//   K6  output projection      scores = h @ fc.W^T   (NT, reference model.py:67)
//   K8  backward data GEMMs    dx = dG @ W  == dG @ (W^T)^T via pre-transposed
//       shadow weights (NT), and weight-grad GEMMs dW = dG^T @ x (TN).
//
// Design: 128x128 output tile, BK=64 K-step, 256 threads = 4 waves in a
// 2x2 wave grid (64x64 per wave = 4x4 fragments of v_mfma_f32_16x16x32_bf16,
// fp32 accumulation). Operands are register-staged into XOR-swizzled LDS
// tiles ([row][k] images, byte ^= (row&7)<<4 — conflict-free-ish for the
// ds_read_b128 column-slice pattern). The TN path transposes at staging
// time with lane-coalesced strided loads. Epilogue fuses the bias add and
// the output dtype cast (bf16 or f32). All edges bounds-guarded with
// zero fill, so arbitrary M/N/K work (vocab 10000 etc.).
#include "common.h"

namespace zamd {

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int GEMM_THREADS = 256;

// LDS image: [128 rows][BK cols] bf16, row stride BK*2 = 128 B, byte
// offset XORed with (row&7)<<4.
DEV_INLINE int swz(int row, int byte_col) {
  return row * (BK * 2) + (byte_col ^ ((row & 7) << 4));
}

// Stage a [rows=128][BK] k-contiguous operand tile: src[row][k] with
// row-major leading dimension ld (elements). Guards both edges, zero fill.
DEV_INLINE void stage_kcontig(const bf16* __restrict__ src, int ld,
                              int row0, int nrows, int k0, int K,
                              bf16* lds) {
  // 256 threads; each pass covers 32 rows x 64 k; 4 passes.
  int t = threadIdx.x;
  int kk = (t & 7) * 8;          // 0..56
  int r = t >> 3;                // 0..31
#pragma unroll
  for (int pass = 0; pass < 4; ++pass) {
    int row = r + pass * 32;
    int gr = row0 + row;
    int gk = k0 + kk;
    bf16x8 v = {};
    if (gr < row0 + nrows && gr >= 0) {
      const bf16* p = src + (int64_t)gr * ld + gk;
      if (gk + 8 <= K) {
        v = *reinterpret_cast<const bf16x8*>(p);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = (gk + e < K) ? p[e] : (bf16)0.f;
      }
    }
    *reinterpret_cast<bf16x8*>(
        reinterpret_cast<char*>(lds) + swz(row, kk * 2)) = v;
  }
}

// Stage a transposed operand: src[K, rows] row-major (k-strided per output
// row). Builds the same [row][k] LDS image. Lanes cover consecutive rows
// so each of the 8 per-k loads is coalesced across the wave.
DEV_INLINE void stage_transpose(const bf16* __restrict__ src, int ld,
                                int row0, int nrows, int k0, int K,
                                bf16* lds) {
  int t = threadIdx.x;
  int row = t & 127;             // output row (= source column)
  int kb = (t >> 7) * 8;         // 0 or 8
  int gr = row0 + row;
  bool rok = row < nrows;
#pragma unroll
  for (int pass = 0; pass < 4; ++pass) {
    int kk = kb + pass * 16;     // 0..56
    int gk = k0 + kk;
    bf16x8 v = {};
    if (rok) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        v[e] = (gk + e < K) ? src[(int64_t)(gk + e) * ld + gr] : (bf16)0.f;
      }
    }
    *reinterpret_cast<bf16x8*>(
        reinterpret_cast<char*>(lds) + swz(row, kk * 2)) = v;
  }
}

DEV_INLINE bf16x8 frag_read(const bf16* lds, int row, int k) {
  return *reinterpret_cast<const bf16x8*>(
      reinterpret_cast<const char*>(lds) + swz(row, k * 2));
}

// C[M,N] = A' @ B' + bias, where A' is A[M,K] (TRANS_A=false) or
// A[K,M] transposed (TRANS_A=true); B' is B[N,K]^T (TRANS_B=false; the
// "NT" weight layout W[N,K] row-major) or B[K,N] (TRANS_B=true staging
// transpose). OutT in {bf16, float}.
template <bool TRANS_A, bool TRANS_B, typename OutT>
__global__ __launch_bounds__(GEMM_THREADS) void gemm_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    OutT* __restrict__ C, const float* __restrict__ bias,
    int M, int N, int K, int lda, int ldb, int ldc) {
  __shared__ bf16 As[BM * BK];
  __shared__ bf16 Bs[BN * BK];

  const int nbn = (N + BN - 1) / BN;
  const int bm = blockIdx.x / nbn;
  const int bn = blockIdx.x % nbn;
  const int m0 = bm * BM, n0 = bn * BN;

  const int w = wave_id();          // 0..3 -> 2x2 wave grid
  const int wm = (w >> 1) * 64;     // wave row offset in tile
  const int wn = (w & 1) * 64;      // wave col offset in tile
  const int l = lane_id();
  const int lm = l & 15;            // fragment row lane
  const int lk = (l >> 4) * 8;      // fragment k offset

  f32x4 acc[4][4] = {};

  for (int k0 = 0; k0 < K; k0 += BK) {
    if (TRANS_A)
      stage_transpose(A, lda, m0, min(BM, M - m0), k0, K, As);
    else
      stage_kcontig(A, lda, m0, min(BM, M - m0), k0, K, As);
    if (TRANS_B)
      stage_transpose(B, ldb, n0, min(BN, N - n0), k0, K, Bs);
    else
      stage_kcontig(B, ldb, n0, min(BN, N - n0), k0, K, Bs);
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      bf16x8 af[4], bfr[4];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        af[i] = frag_read(As, wm + i * 16 + lm, kk + lk);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bfr[j] = frag_read(Bs, wn + j * 16 + lm, kk + lk);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = mfma_16x16x32_bf16(af[i], bfr[j], acc[i][j]);
    }
    __syncthreads();
  }

  // Epilogue: C/D fragment map for 16x16: col = l&15, row = (l>>4)*4 + r.
  const int fc_ = l & 15;
  const int fr0 = (l >> 4) * 4;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int col = n0 + wn + j * 16 + fc_;
    if (col >= N) continue;
    float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wm + i * 16 + fr0 + r;
        if (row < M) {
          float v = acc[i][j][r] + bv;
          C[(int64_t)row * ldc + col] = (OutT)v;
        }
      }
    }
  }
}

template <bool TA, bool TB, typename OutT>
void launch_gemm_t(const bf16* A, const bf16* B, OutT* C, const float* bias,
                   int M, int N, int K, int lda, int ldb, int ldc,
                   hipStream_t stream) {
  int grid = cdiv(M, BM) * cdiv(N, BN);
  hipLaunchKernelGGL((gemm_kernel<TA, TB, OutT>), dim3(grid),
                     dim3(GEMM_THREADS), 0, stream, A, B, C, bias, M, N, K,
                     lda, ldb, ldc);
}

// Explicit instantiations used by ext.cpp
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
