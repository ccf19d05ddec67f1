// MFMA bf16 GEMM kernels for gfx950 (CDNA4).
//
// Covers the framework's dense-GEMM call sites:
//   K2  input-side gate GEMM   gx = x @ W_x^T        (NT, reference model.py:35)
//   K6  output projection      scores = h @ fc.W^T   (NT, reference model.py:67)
//   K8  backward data GEMMs    dx = dG @ W (NT via pre-transposed,
//       K-zero-padded shadow weights; 2-way split-K on the ~1-block/CU
//       shapes) and weight-grad GEMMs dW = dG^T @ x (as transpose + NT
//       over K-padded temps; the TN staging path exists but is
//       register-starved and unused on the hot path).
//
// Design notes (measured on MI355X, see profiles/):
//   * TMxTN output tile with a WGMxWGN wave grid. Shapes pick their
//     tile: 256x128 with 8 waves for the big dW/projection GEMMs
//     (double arithmetic intensity + twice the staging streams),
//     128x128 / 64x64 with 4 waves for mid/skinny shapes so the grid
//     still fills the 256 CUs, 128x64 for the split-K dx family.
//   * BK=64, v_mfma_f32_16x16x32_bf16 with fp32 accumulation.
//   * Register-staged LDS with the T14 async-split schedule: issue tile
//     t+1's global loads immediately after publishing tile t to LDS, so
//     HBM/L2 latency hides under the MFMA phase.
//   * Interior blocks take a guard-free staging path — per-element bounds
//     branches around global loads de-pipeline hipcc's vmcnt bookkeeping
//     (one drained load per element), measured 3-5x slower.
//   * LDS images are [row][k] with byte ^= (row&7)<<4 XOR swizzle:
//     conflict-reduced ds_read_b128 column-slice reads.
//   * The TN (transposed) staging loads k-rows per thread with
//     lane-coalesced strided reads and writes one b128 per thread.
#include "common.h"

namespace zamd {

constexpr int BK = 64;

template <int BROWS>
DEV_INLINE int swz(int row, int byte_col) {
  return row * (BK * 2) + (byte_col ^ ((row & 7) << 4));
}

// ---- async global->LDS staging (interior fast path) -----------------------
// One glds wave-instruction moves 1 KB: 8 rows x 128 B, lane l -> LDS byte
// (wave-uniform base) + l*16. The LDS image stays linear; the XOR swizzle
// moves to the per-lane SOURCE address and the matching XOR on the read
// side (guide §5.4 rule 21: same involution on source and read).
DEV_INLINE void glds16(const bf16* gsrc, bf16* lds_dst_uniform) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)gsrc,
      (__attribute__((address_space(3))) void*)lds_dst_uniform, 16, 0, 0);
}

// Stage a [TILE][BK] k-contiguous tile via glds: TILE*BK*2/1024
// instructions split across NWAVES waves. Out-of-range rows are CLAMPED
// to the last valid row: the garbage values land only in output
// rows/cols the epilogue discards (each output element depends only on
// its own A row and B col), so M/N-edge blocks ride the same fast path —
// only K-tail tiles need the guarded register staging (tail columns feed
// REAL outputs and must be zero).
template <int TILE, int NWAVES>
DEV_INLINE void stage_glds_kc(const bf16* __restrict__ src, int ld, int row0,
                              int nrows_total, int k0, bf16* lds) {
  constexpr int NINST = TILE * BK * 2 / 1024;  // 1 KB per instruction
  constexpr int PER_WAVE = (NINST + NWAVES - 1) / NWAVES;
  const int w = wave_id();
  const int l = lane_id();
#pragma unroll
  for (int i = 0; i < PER_WAVE; ++i) {
    const int inst = w * PER_WAVE + i;
    if (inst >= NINST) break;
    const int row = inst * 8 + (l >> 3);          // 8 rows per instruction
    const int gr = min(row0 + row, nrows_total - 1);
    const int colb = (l & 7) * 16;                // byte column 0..112
    const int src_colb = colb ^ ((row & 7) << 4); // pre-swizzled source
    glds16(src + (int64_t)gr * ld + k0 + src_colb / 2,
           lds + (int64_t)inst * 512);            // 1024 B = 512 bf16
  }
}

// ---- staging: k-contiguous operand (NT layout), [BROWS][BK] tile ---------
// Register-load phase: each thread grabs BROWS*BK/(THREADS*8) bf16x8
// vectors.
template <int BROWS, int THREADS, bool GUARD>
DEV_INLINE void stage_load_kc(const bf16* __restrict__ src, int ld, int row0,
                              int nrows, int k0, int K,
                              bf16x8 (&v)[BROWS * 8 / THREADS]) {
  const int t = threadIdx.x;
  const int kk = (t & 7) * 8;
  const int r = t >> 3;                 // 0 .. THREADS/8-1
  constexpr int RSTRIDE = THREADS / 8;  // rows per pass
#pragma unroll
  for (int p = 0; p < BROWS * 8 / THREADS; ++p) {
    const int row = r + p * RSTRIDE;
    if (GUARD) {
      // Branch-free guards: loads always execute from CLAMPED in-bounds
      // addresses; out-of-range values are zeroed by VALUE selects.
      // (Branching around loads makes hipcc drain vmcnt per element —
      // measured ~12 us per K-tail tile before.) Row clamp alone is
      // enough for the M/N direction: pad-row garbage only reaches
      // discarded outputs; the K direction must read as ZERO.
      // all-scalar, all-unconditional: even a per-thread branch around a
      // vector load serializes hipcc's waitcnt bookkeeping
      const int gr = row0 + min(row, nrows - 1);
      const int gk = k0 + kk;
      const bf16* sp = src + (int64_t)gr * ld;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        bf16 x = sp[min(gk + e, K - 1)];
        v[p][e] = (gk + e < K) ? x : (bf16)0.f;
      }
    } else {
      v[p] = *reinterpret_cast<const bf16x8*>(src + (int64_t)(row0 + row) * ld +
                                              k0 + kk);
    }
  }
}

template <int BROWS, int THREADS>
DEV_INLINE void stage_write_kc(bf16* lds, bf16x8 (&v)[BROWS * 8 / THREADS]) {
  const int t = threadIdx.x;
  const int kk = (t & 7) * 8;
  const int r = t >> 3;
  constexpr int RSTRIDE = THREADS / 8;
#pragma unroll
  for (int p = 0; p < BROWS * 8 / THREADS; ++p)
    *reinterpret_cast<bf16x8*>(reinterpret_cast<char*>(lds) +
                               swz<BROWS>(r + p * RSTRIDE, kk * 2)) = v[p];
}

// ---- staging: transposed operand (TN layout: src[K, rows]) ----------------
template <int BROWS, int THREADS, bool GUARD>
DEV_INLINE void stage_load_tr(const bf16* __restrict__ src, int ld, int row0,
                              int nrows, int k0, int K,
                              bf16x8 (&v)[BROWS * 8 / THREADS]) {
  const int t = threadIdx.x;
  const int row = t & (BROWS - 1);
  const int kb = (t / BROWS) * 8;           // THREADS/BROWS k-groups/pass
  constexpr int KG = 8 * (THREADS / BROWS); // k covered per pass
  const int gr = row0 + row;
#pragma unroll
  for (int p = 0; p < BROWS * 8 / THREADS; ++p) {  // == BK/KG passes
    const int kk = kb + p * KG;
    if (GUARD) {
      // clamped-address loads + value selects (see stage_load_kc note)
      const int grc = row0 + min(row, nrows - 1);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int gk = k0 + kk + e;
        bf16 x = src[(int64_t)min(gk, K - 1) * ld + grc];
        v[p][e] = (gk < K) ? x : (bf16)0.f;
      }
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e)
        v[p][e] = src[(int64_t)(k0 + kk + e) * ld + gr];
    }
  }
}

template <int BROWS, int THREADS>
DEV_INLINE void stage_write_tr(bf16* lds, bf16x8 (&v)[BROWS * 8 / THREADS]) {
  const int t = threadIdx.x;
  const int row = t & (BROWS - 1);
  const int kb = (t / BROWS) * 8;
  constexpr int KG = 8 * (THREADS / BROWS);
#pragma unroll
  for (int p = 0; p < BROWS * 8 / THREADS; ++p)
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
// otherwise the k-contiguous layouts A[M,K] / B[N,K]. TMxTN tile,
// WGMxWGN wave grid (THREADS = WGM*WGN*64), each wave (TM/WGM)x(TN/WGN).
// ---------------------------------------------------------------------------
template <int TM, int TN, int WGM, int WGN, bool TRANS_A, bool TRANS_B,
          typename OutT>
__global__ __launch_bounds__(WGM* WGN * 64) void gemm_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    OutT* __restrict__ C, const float* __restrict__ bias, int M, int N,
    int K, int lda, int ldb, int ldc, OutT* __restrict__ C2, int kt_split) {
  // N-way split-K (kt_split > 0): blockIdx.z picks a K-tile range and a
  // partial output (z0 -> C with bias; z>0 -> slice z-1 of the stacked
  // C2 partials, no bias). gridDim.z co-resident blocks/CU interleave
  // their latency chains on the ~1-block/CU dx shapes (measured:
  // doubling the grid is near-free). Caller sums C + C2 slices.
  if (kt_split > 0) {
    const int z = blockIdx.z;
    if (z) {
      const int64_t koff = (int64_t)kt_split * BK * z;
      A += koff;
      B += koff;
      K -= (int)(kt_split * BK * z);
      C = C2 + (int64_t)(z - 1) * M * ldc;
      bias = nullptr;
    }
    if (z + 1 < (int)gridDim.z) K = kt_split * BK;
  }
  constexpr int THREADS = WGM * WGN * 64;
  constexpr int WTM = TM / WGM;      // wave tile rows
  constexpr int WTN = TN / WGN;      // wave tile cols
  constexpr int NFM = WTM / 16;      // row fragments per wave
  constexpr int NFN = WTN / 16;      // col fragments per wave

  // XCD-aware tile assignment: the dispatcher round-robins blockIdx
  // across the 8 XCDs, so consecutive tiles (which share an A row-slice
  // under the n-major tile order) land on DIFFERENT XCDs and every
  // private L2 re-pulls nearly the whole operand set from LLC. Remap so
  // each XCD owns a CONTIGUOUS tile stripe (a bijection handling the
  // ragged tail: the first nblk%8 XCDs own ceil stripes).
  int tile_id = blockIdx.x;
  {
    const int nblk = gridDim.x;
    const int xcd = blockIdx.x & 7;
    const int loc = blockIdx.x >> 3;
    const int per_lo = nblk >> 3, r = nblk & 7;
    tile_id = (xcd < r) ? xcd * (per_lo + 1) + loc
                        : r * (per_lo + 1) + (xcd - r) * per_lo + loc;
  }
  // one LDS object only (a second __shared__ forces vmcnt(0) before every
  // ds_read when a glds is in flight — guide §5 trap (a)); double-buffered.
  // Dynamic LDS: the 256x128 tile needs 96 KB, past the 64 KB static
  // limit (gfx950 allows 160 KB/WG dynamically).
  extern __shared__ __attribute__((aligned(16))) char gemm_smem[];
  bf16* As = reinterpret_cast<bf16*>(gemm_smem);  // [2][TM*BK] A buffers
  bf16* Bs = As + 2 * TM * BK;

  const int nbn = (N + TN - 1) / TN;
  const int bm = tile_id / nbn;
  const int bn = tile_id % nbn;
  const int m0 = bm * TM, n0 = bn * TN;
  const bool interior =
      (m0 + TM <= M) && (n0 + TN <= N);

  const int w = wave_id();
  const int wm = (w / WGN) * WTM;
  const int wn = (w % WGN) * WTN;
  const int l = lane_id();
  const int lm = l & 15;
  const int lk = (l >> 4) * 8;

  f32x4 acc[NFM][NFN] = {};
  bf16x8 va[TM * 8 / THREADS], vb[TN * 8 / THREADS];

  const int nk = (K + BK - 1) / BK;
  const int k_full = K / BK;         // tiles with no K/MN guard

  auto load_tile = [&](int kt, bool guard_mn) {
    const bool gk = (kt >= k_full);
    if (!guard_mn && !gk) {
      if (TRANS_A)
        stage_load_tr<TM, THREADS, false>(A, lda, m0, TM, kt * BK, K, va);
      else
        stage_load_kc<TM, THREADS, false>(A, lda, m0, TM, kt * BK, K, va);
      if (TRANS_B)
        stage_load_tr<TN, THREADS, false>(B, ldb, n0, TN, kt * BK, K, vb);
      else
        stage_load_kc<TN, THREADS, false>(B, ldb, n0, TN, kt * BK, K, vb);
    } else {
      if (TRANS_A)
        stage_load_tr<TM, THREADS, true>(A, lda, m0, min(TM, M - m0), kt * BK,
                                         K, va);
      else
        stage_load_kc<TM, THREADS, true>(A, lda, m0, min(TM, M - m0), kt * BK,
                                         K, va);
      if (TRANS_B)
        stage_load_tr<TN, THREADS, true>(B, ldb, n0, min(TN, N - n0), kt * BK,
                                         K, vb);
      else
        stage_load_kc<TN, THREADS, true>(B, ldb, n0, min(TN, N - n0), kt * BK,
                                         K, vb);
    }
  };

  auto mfma_phase = [&](const bf16* Ab, const bf16* Bb) {
#pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      bf16x8 af[NFM], bfr[NFN];
#pragma unroll
      for (int i = 0; i < NFM; ++i)
        af[i] = frag_read<TM>(Ab, wm + i * 16 + lm, kk + lk);
#pragma unroll
      for (int j = 0; j < NFN; ++j)
        bfr[j] = frag_read<TN>(Bb, wn + j * 16 + lm, kk + lk);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < NFM; ++i)
#pragma unroll
        for (int j = 0; j < NFN; ++j)
          acc[i][j] = mfma_16x16x32_bf16(af[i], bfr[j], acc[i][j]);
      __builtin_amdgcn_s_setprio(0);
    }
  };

  if (!TRANS_A && !TRANS_B && k_full > 0) {
    // glds 2-phase pipeline (guide §5.5 T3 minimum form): stage tile t+1
    // while computing tile t; __syncthreads() drains the in-flight glds.
    // Edge blocks use row-clamped sources (see stage_glds_kc).
    constexpr int NWAVES = WGM * WGN;
    stage_glds_kc<TM, NWAVES>(A, lda, m0, M, 0, As);
    stage_glds_kc<TN, NWAVES>(B, ldb, n0, N, 0, Bs);
    __syncthreads();
    int cur = 0;
    for (int kt = 0; kt < k_full; ++kt) {
      if (kt + 1 < k_full) {
        stage_glds_kc<TM, NWAVES>(A, lda, m0, M, (kt + 1) * BK,
                                  As + (cur ^ 1) * TM * BK);
        stage_glds_kc<TN, NWAVES>(B, ldb, n0, N, (kt + 1) * BK,
                                  Bs + (cur ^ 1) * TN * BK);
      }
      mfma_phase(As + cur * TM * BK, Bs + cur * TN * BK);
      __syncthreads();
      cur ^= 1;
    }
    if (k_full < nk) {  // K tail: register-staged, guarded
      stage_load_kc<TM, THREADS, true>(A, lda, m0, min(TM, M - m0),
                                       k_full * BK, K, va);
      stage_load_kc<TN, THREADS, true>(B, ldb, n0, min(TN, N - n0),
                                       k_full * BK, K, vb);
      stage_write_kc<TM, THREADS>(As + cur * TM * BK, va);
      stage_write_kc<TN, THREADS>(Bs + cur * TN * BK, vb);
      __syncthreads();
      mfma_phase(As + cur * TM * BK, Bs + cur * TN * BK);
    }
  } else {
    // register-staged path (edge blocks / transposed operands)
    constexpr bool ASYNC = !(TRANS_A || TRANS_B);
    if (ASYNC) load_tile(0, !interior);
    for (int kt = 0; kt < nk; ++kt) {
      if (!ASYNC) load_tile(kt, !interior);
      __syncthreads();  // LDS consumers of tile kt-1 done
      if (TRANS_A)
        stage_write_tr<TM, THREADS>(As, va);
      else
        stage_write_kc<TM, THREADS>(As, va);
      if (TRANS_B)
        stage_write_tr<TN, THREADS>(Bs, vb);
      else
        stage_write_kc<TN, THREADS>(Bs, vb);
      if (ASYNC && kt + 1 < nk) load_tile(kt + 1, !interior);  // issue early
      __syncthreads();
      mfma_phase(As, Bs);
    }
  }

  // Epilogue. C/D map: col = l&15, row = (l>>4)*4 + r.
  const int fr0 = (l >> 4) * 4;
#pragma unroll
  for (int j = 0; j < NFN; ++j) {
    const int col = n0 + wn + j * 16 + (l & 15);
    if (col >= N) continue;
    const float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int i = 0; i < NFM; ++i) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm + i * 16 + fr0 + r;
        if (row < M) C[(int64_t)row * ldc + col] = (OutT)(acc[i][j][r] + bv);
      }
    }
  }
}

// ZAMD_GEMM_TILE forces a tile for A/B sweeps: 256 (256x128/8w),
// 128 (128^2/4w), 12864 (128x64/4w), 64 (64^2/4w); unset/0 = heuristic.
static int force_tile() {
  static int v = [] {
    const char* e = getenv("ZAMD_GEMM_TILE");
    return e ? atoi(e) : 0;
  }();
  return v;
}

template <bool TA, bool TB, typename OutT>
void launch_gemm_t(const bf16* A, const bf16* B, OutT* C, const float* bias,
                   int M, int N, int K, int lda, int ldb, int ldc,
                   hipStream_t stream) {
  // Tile choice is about CO-RESIDENT BLOCKS PER CU, not arithmetic
  // intensity: probe data (profiles/s7_gemm_probe.txt) shows TF rising
  // with grid size at constant shape family — these mid-size GEMMs are
  // latency-bound. 256x128 (96 KB LDS -> 1 block/CU) measured no better
  // than 128^2 (64 KB -> 2/CU); smaller tiles trade L2-dedup'd re-reads
  // for more interleaved latency chains. The TN path keeps 4-wave tiles
  // (register-starved staging).
  const int f = force_tile();
#define ZAMD_G_LAUNCH(TM_, TN_, WGM_, WGN_)                                   \
  hipLaunchKernelGGL((gemm_kernel<TM_, TN_, WGM_, WGN_, TA, TB, OutT>),       \
                     dim3(cdiv(M, TM_) * cdiv(N, TN_)),                       \
                     dim3(WGM_ * WGN_ * 64), 2 * (TM_ + TN_) * BK * 2,        \
                     stream, A, B, C, bias, M, N, K, lda, ldb, ldc,           \
                     (OutT*)nullptr, 0)
  if (!TA && !TB) {
    if (f == 256) { ZAMD_G_LAUNCH(256, 128, 4, 2); return; }
    if (f == 12864) { ZAMD_G_LAUNCH(128, 64, 2, 2); return; }
    if (f == 128) { ZAMD_G_LAUNCH(128, 128, 2, 2); return; }
    if (f == 64) { ZAMD_G_LAUNCH(64, 64, 2, 2); return; }
    // Measured per-shape-family choices (profiles/s8_sweep.txt):
    //  * tall-skinny dW gate shapes ([6000,1500,768]): 128x64 at ~4
    //    blocks/CU, 32.3 vs 38.6 us on 128^2 (the >8192-M proj-dW
    //    [10000,1500] flips back: 43.2 on 128^2 vs 45.5),
    //  * wide-N forward shapes (proj [700,10000,1500] 46.9 vs 52.6;
    //    input [700,6000,1500] 42.7 vs 45.8): 256x128/8-wave.
    if (M >= 4 * N && M <= 8192 && cdiv(M, 128) * cdiv(N, 64) >= 384) {
      ZAMD_G_LAUNCH(128, 64, 2, 2);
      return;
    }
    if (N >= 2 * M && cdiv(M, 256) * cdiv(N, 128) >= 128) {
      ZAMD_G_LAUNCH(256, 128, 4, 2);
      return;
    }
  }
  int grid128 = cdiv(M, 128) * cdiv(N, 128);
  if (grid128 >= 192) {
    ZAMD_G_LAUNCH(128, 128, 2, 2);
  } else {
    ZAMD_G_LAUNCH(64, 64, 2, 2);
  }
#undef ZAMD_G_LAUNCH
}

// 2-way split-K NT GEMM writing partials C/C2 (caller combines). K must
// be a BK multiple with >= 2 tiles (the k_pad contract guarantees it on
// the dx shapes). 128x64 tile: the [700,1500] dx shapes give a 144-block
// grid, x2 K-split = 288 (~1.1 blocks/CU) with half the B re-reads of
// the old 64^2 (which ran 528 blocks re-pulling the W shadow per tile).
// Returns the split ways used (the caller sums that many partials).
template <typename OutT>
int launch_gemm_splitk_t(const bf16* A, const bf16* B, OutT* C, OutT* C2,
                         const float* bias, int M, int N, int K, int lda,
                         int ldb, int ldc, hipStream_t stream) {
  // 2-way default: the same-box A/B matrix (profiles/s17_ab.txt)
  // measured 4-way SLOWER end-to-end (321.7K vs 319.4K tokens/s) —
  // the shorter K chains don't pay for the extra partial traffic +
  // combine reads on these shapes. ZAMD_SPLITK_NZ=4 for A/B. 64^2
  // tile (tiny blocks interleave latency chains harder than 128x64:
  // 50.7 vs 54.8 us; ZAMD_GEMM_TILE=12864 flips it).
  static const int force_nz = [] {
    const char* e = getenv("ZAMD_SPLITK_NZ");
    return e ? atoi(e) : 0;
  }();
  const int nz = force_nz ? force_nz : 2;
  int kt_split = (K / BK) / nz;
  if (force_tile() == 12864) {
    int grid12864 = cdiv(M, 128) * cdiv(N, 64);
    hipLaunchKernelGGL((gemm_kernel<128, 64, 2, 2, false, false, OutT>),
                       dim3(grid12864, 1, nz), dim3(256),
                       2 * (128 + 64) * BK * 2, stream, A, B, C,
                       bias, M, N, K, lda, ldb, ldc, C2, kt_split);
    return nz;
  }
  int grid64 = cdiv(M, 64) * cdiv(N, 64);
  hipLaunchKernelGGL((gemm_kernel<64, 64, 2, 2, false, false, OutT>),
                     dim3(grid64, 1, nz), dim3(256), 2 * (64 + 64) * BK * 2,
                     stream, A, B, C, bias,
                     M, N, K, lda, ldb, ldc, C2, kt_split);
  return nz;
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
template int launch_gemm_splitk_t<float>(const bf16*, const bf16*, float*,
                                         float*, const float*, int, int,
                                         int, int, int, int, hipStream_t);
template int launch_gemm_splitk_t<bf16>(const bf16*, const bf16*, bf16*,
                                        bf16*, const float*, int, int, int,
                                        int, int, int, hipStream_t);

}  // namespace zamd
