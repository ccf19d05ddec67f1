// Fused LSTM cell kernels for gfx950 — the sequential hot path
// (SURVEY.md §2.3 K3+K4, reference model.py:34-45 math; gate order i,f,o,n).
//
// Forward, one timestep (launched T times per layer by the C++ sequence
// driver in ext_bind.hip, hipGraph-captured):
//   gates = h_prev @ W_h^T + gx_t          (gx already holds x@W_x^T+b_x+b_h)
//   i,f,o = sigmoid(g0,g1,g2); n = tanh(g3)
//   c = f*c_prev + i*n ; h = o*tanh(c)
//
// Fragment-packed operand design (MI355X-specific, measured):
//   The natural MFMA fragment access for a skinny [B<=32, H] x [H, 4H]
//   GEMM scatters every 16-B lane load across 16 weight rows — each
//   wave instruction becomes ~16 separate 64-B L2 transactions, and the
//   kernel is transaction-rate-bound (~17 us/step). Instead, both
//   operands are kept in FRAGMENT-PACKED layout ([kstep][frag][lane][8]
//   bf16), so every wave load is one contiguous 1 KB burst:
//     * W_h / W_h^T are packed once per SGD step (pack_gated_w kernel,
//       reading the bf16 shadows) — amortized over the 70 cell launches
//       of every training step,
//     * h is written in packed layout by the PREVIOUS cell step's
//       epilogue (dual store), and h0 by a tiny pack_a kernel at
//       sequence start,
//     * dgates are written packed by the backward elementwise kernel for
//       the recurrent-hop GEMM that follows it.
//   Pad rows (b >= B) and K-tail slots are zero in the packed buffers
//   (zero-prefilled workspaces / zero-filled packing), so the inner
//   loops are completely uniform — no bounds branches, no frag guards.
//
// Packed layout: frag f of kstep ks, lane l, elem e lives at
//   ((ks*2 + f)*64 + l)*8 + e          (A operands: 2 M-fragments)
//   (((blk*NG + g)*KS + ks)*64 + l)*8 + e   (W operands, NG gates)
// with the standard v_mfma_f32_16x16x32_bf16 maps
//   A: row = (l&15) + 16*f, k = ks*32 + (l>>4)*8 + e
//   B: col = l&15,          k = ks*32 + (l>>4)*8 + e.
#include "common.h"

namespace zamd {

constexpr int CELL_THREADS = 256;

// ---------------------------------------------------------------------------
// Packing kernels
// ---------------------------------------------------------------------------
// W[N-space, K] (row-major, k-contiguous; N-space = ngates * rows) ->
// packed [ceil(rows/16)][ngates][ceil(K/32)][64][8], zero-filled pads.
__global__ void pack_gated_w_kernel(const bf16* __restrict__ W,
                                    bf16* __restrict__ out, int rows,
                                    int ngates, int K) {
  const int KS = (K + 31) / 32;
  const int nb = (rows + 15) / 16;
  const int64_t total = (int64_t)nb * ngates * KS * 64;
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= total) return;
  const int l = idx & 63;
  const int ks = (idx >> 6) % KS;
  const int g = (int)((idx >> 6) / KS) % ngates;
  const int bj = (int)((idx >> 6) / KS / ngates);
  const int row = bj * 16 + (l & 15);
  const int k = ks * 32 + (l >> 4) * 8;
  bf16x8 v = {};
  if (row < rows) {
    const bf16* p = W + ((int64_t)g * rows + row) * K + k;
    if (k + 8 <= K) {
      v = *reinterpret_cast<const bf16x8*>(p);
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) v[e] = (k + e < K) ? p[e] : (bf16)0.f;
    }
  }
  reinterpret_cast<bf16x8*>(out)[idx] = v;
}

void launch_pack_gated_w(const bf16* W, bf16* out, int rows, int ngates,
                         int K, hipStream_t stream) {
  const int KS = (K + 31) / 32;
  const int nb = (rows + 15) / 16;
  int64_t total = (int64_t)nb * ngates * KS * 64;
  hipLaunchKernelGGL(pack_gated_w_kernel, dim3(cdiv(total, 256)), dim3(256),
                     0, stream, W, out, rows, ngates, K);
}

// A[B, K] -> packed [ceil(K/32)][2][64][8]; pad rows/K-tail slots must
// already be zero in `out` (zero-prefilled persistent workspace).
__global__ void pack_a_kernel(const bf16* __restrict__ A,
                              bf16* __restrict__ out, int B, int K) {
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (int64_t)B * K) return;
  const int b = (int)(idx / K);
  const int k = (int)(idx % K);
  const int ks = k / 32, sub = k % 32;
  const int l = (b & 15) + 16 * (sub / 8);
  const int e = sub % 8;
  const int f = b / 16;
  out[(((int64_t)ks * 2 + f) * 64 + l) * 8 + e] = A[idx];
}

void launch_pack_a(const bf16* A, bf16* out, int B, int K,
                   hipStream_t stream) {
  hipLaunchKernelGGL(pack_a_kernel, dim3(cdiv((int64_t)B * K, 256)),
                     dim3(256), 0, stream, A, out, B, K);
}

// ---------------------------------------------------------------------------
// Forward cell (packed operands)
// ---------------------------------------------------------------------------
template <int MAXB>
__global__ __launch_bounds__(CELL_THREADS) void lstm_cell_fwd_kernel(
    const bf16* __restrict__ h_pack,   // [KS][2][64][8] packed h_{t}
    const float* __restrict__ c_prev,  // [B, H]
    const bf16* __restrict__ gx,       // [B, 4H] this timestep's input gates
    const bf16* __restrict__ W_pack,   // [nb][4][KS][64][8]
    bf16* __restrict__ h_out,          // [B, H]
    bf16* __restrict__ h_pack_out,     // packed h_{t+1}
    float* __restrict__ c_out,         // [B, H]
    bf16* __restrict__ gates_out,      // [B, 4H] post-activation i,f,o,n
    bf16* __restrict__ rec,            // [NB][B][6][HSp] block record for
                                       // this t (persistent_hs layout)
    int B, int H, int HSp) {
  __shared__ float gbuf[4 * MAXB * 16];

  const int j0 = blockIdx.x * 16;  // hidden-unit slice
  const int g = wave_id();         // gate index (i,f,o,n)
  const int l = lane_id();
  const int lm = l & 15;
  const int KS = (H + 31) / 32;

  const bf16x8* pa = reinterpret_cast<const bf16x8*>(h_pack) + l;
  const bf16x8* pw = reinterpret_cast<const bf16x8*>(W_pack) +
                     ((int64_t)blockIdx.x * 4 + g) * KS * 64 + l;

  f32x4 acc0 = {}, acc1 = {};
  int ks = 0;
  for (; ks + 8 <= KS; ks += 8) {
    bf16x8 a0v[8], a1v[8], bwv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      a0v[u] = pa[(ks + u) * 128];
      a1v[u] = pa[(ks + u) * 128 + 64];
      bwv[u] = pw[(ks + u) * 64];
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      acc0 = mfma_16x16x32_bf16(a0v[u], bwv[u], acc0);
      acc1 = mfma_16x16x32_bf16(a1v[u], bwv[u], acc1);
    }
  }
  for (; ks < KS; ++ks) {
    bf16x8 a0v = pa[ks * 128];
    bf16x8 a1v = pa[ks * 128 + 64];
    bf16x8 bwv = pw[ks * 64];
    acc0 = mfma_16x16x32_bf16(a0v, bwv, acc0);
    acc1 = mfma_16x16x32_bf16(a1v, bwv, acc1);
  }

  // ---- exchange gate tiles through LDS -----------------------------------
  // C/D map: col = l&15, row = (l>>4)*4 + r.
  const int fr0 = (l >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    gbuf[(g * MAXB + fr0 + r) * 16 + lm] = acc0[r];
    gbuf[(g * MAXB + 16 + fr0 + r) * 16 + lm] = acc1[r];
  }
  __syncthreads();

  // ---- pointwise cell update (dual h store: row-major + packed) ----------
  for (int idx = threadIdx.x; idx < B * 16; idx += CELL_THREADS) {
    const int b = idx / 16;
    const int jj = idx % 16;
    const int j = j0 + jj;
    if (j >= H) continue;
    const int64_t gbase = (int64_t)b * 4 * H + j;
    float gi = gbuf[(0 * MAXB + b) * 16 + jj] + bf2f(gx[gbase + 0 * H]);
    float gf = gbuf[(1 * MAXB + b) * 16 + jj] + bf2f(gx[gbase + 1 * H]);
    float go = gbuf[(2 * MAXB + b) * 16 + jj] + bf2f(gx[gbase + 2 * H]);
    float gn = gbuf[(3 * MAXB + b) * 16 + jj] + bf2f(gx[gbase + 3 * H]);
    float i_ = 1.f / (1.f + __expf(-gi));
    float f_ = 1.f / (1.f + __expf(-gf));
    float o_ = 1.f / (1.f + __expf(-go));
    float n_ = tanhf(gn);
    const float cp = c_prev[(int64_t)b * H + j];
    float c_ = f_ * cp + i_ * n_;
    const float tc = tanhf(c_);
    float h_ = o_ * tc;
    const bf16 hb = f2bf(h_);
    c_out[(int64_t)b * H + j] = c_;
    h_out[(int64_t)b * H + j] = hb;
    {  // block record in the persistent layout (backward reads this)
      const int blk = j / HSp, jr = j % HSp;
      bf16* r = rec + (((int64_t)blk * B + b) * 6) * HSp;
      r[0 * HSp + jr] = f2bf(i_);
      r[1 * HSp + jr] = f2bf(f_);
      r[2 * HSp + jr] = f2bf(o_);
      r[3 * HSp + jr] = f2bf(n_);
      r[4 * HSp + jr] = f2bf(tc);
      r[5 * HSp + jr] = f2bf(cp);
    }
    {  // packed slot for the next step's A operand
      const int ks2 = j / 32, sub = j % 32;
      const int pl = (b & 15) + 16 * (sub / 8);
      h_pack_out[(((int64_t)ks2 * 2 + b / 16) * 64 + pl) * 8 + sub % 8] = hb;
    }
    gates_out[gbase + 0 * H] = f2bf(i_);
    gates_out[gbase + 1 * H] = f2bf(f_);
    gates_out[gbase + 2 * H] = f2bf(o_);
    gates_out[gbase + 3 * H] = f2bf(n_);
  }
}

void launch_lstm_cell_fwd(const bf16* h_pack, const float* c_prev,
                          const bf16* gx, const bf16* W_pack, bf16* h_out,
                          bf16* h_pack_out, float* c_out, bf16* gates_out,
                          bf16* rec, int B, int H, int HSp,
                          hipStream_t stream) {
  hipLaunchKernelGGL((lstm_cell_fwd_kernel<32>), dim3(cdiv(H, 16)),
                     dim3(CELL_THREADS), 0, stream, h_pack, c_prev, gx,
                     W_pack, h_out, h_pack_out, c_out, gates_out, rec, B, H,
                     HSp);
}

// ---------------------------------------------------------------------------
// Backward: per-timestep elementwise dgate kernel (dual dG store)
// ---------------------------------------------------------------------------
__global__ void lstm_cell_bwd_elt_kernel(
    const bf16* __restrict__ dy, const float* __restrict__ dh_rec,
    const float* __restrict__ dh_rec2, float* __restrict__ dc,
    const bf16* __restrict__ rec, bf16* __restrict__ dG,
    bf16* __restrict__ dG_pack, int B, int H, int HSp) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= B * H) return;
  int b = idx / H, j = idx % H;
  const int64_t gbase = (int64_t)b * 4 * H + j;
  const int blk = j / HSp, jr = j % HSp;
  const bf16* r = rec + (((int64_t)blk * B + b) * 6) * HSp;
  float i_ = bf2f(r[0 * HSp + jr]);
  float f_ = bf2f(r[1 * HSp + jr]);
  float o_ = bf2f(r[2 * HSp + jr]);
  float n_ = bf2f(r[3 * HSp + jr]);
  float tc = bf2f(r[4 * HSp + jr]);
  float cprev = bf2f(r[5 * HSp + jr]);
  // first reverse step (dh_rec == null): carried state starts at zero
  float dh = bf2f(dy[idx]) + (dh_rec ? dh_rec[idx] + dh_rec2[idx] : 0.f);
  float do_ = dh * tc;
  float dct = (dh_rec ? dc[idx] : 0.f) + dh * o_ * (1.f - tc * tc);
  float di = dct * n_;
  float df = dct * cprev;
  float dn = dct * i_;
  const bf16 v[4] = {f2bf(di * i_ * (1.f - i_)), f2bf(df * f_ * (1.f - f_)),
                     f2bf(do_ * o_ * (1.f - o_)), f2bf(dn * (1.f - n_ * n_))};
#pragma unroll
  for (int g = 0; g < 4; ++g) {
    const int k = g * H + j;  // column in [B, 4H]
    dG[gbase + (int64_t)g * H] = v[g];
    const int ks = k / 32, sub = k % 32;
    const int pl = (b & 15) + 16 * (sub / 8);
    dG_pack[(((int64_t)ks * 2 + b / 16) * 64 + pl) * 8 + sub % 8] = v[g];
  }
  dc[idx] = dct * f_;
}

void launch_lstm_cell_bwd_elt(const bf16* dy, const float* dh_rec,
                              const float* dh_rec2, float* dc,
                              const bf16* rec, bf16* dG, bf16* dG_pack,
                              int B, int H, int HSp, hipStream_t stream) {
  int n = B * H;
  hipLaunchKernelGGL(lstm_cell_bwd_elt_kernel, dim3(cdiv(n, 256)), dim3(256),
                     0, stream, dy, dh_rec, dh_rec2, dc, rec, dG, dG_pack, B,
                     H, HSp);
}

// ---------------------------------------------------------------------------
// Skinny-M NT GEMM, packed operands: C[M<=32, N] f32 = A @ B^T
// ---------------------------------------------------------------------------
// The recurrent backward hop dh_rec = dG_t @ W_h. A_pack is the packed
// dG ([KS][2][64][8]); W_pack is the packed W_h^T shadow
// ([ceil(N/16)][KS][64][8]). Grid = ceil(N/16); the 4 waves round-robin
// the K steps; one LDS reduction at the end. Fully uniform — pads are
// zero in both packs.
template <int MAXB, int SK>
__global__ __launch_bounds__(CELL_THREADS) void smallm_packed_nt_kernel(
    const bf16* __restrict__ A_pack, const bf16* __restrict__ W_pack,
    float* __restrict__ C, float* __restrict__ C2, int M, int N, int K) {
  __shared__ float red[4 * MAXB * 16];

  const int nbn = (N + 15) / 16;
  const int n0 = (blockIdx.x % nbn) * 16;
  const int sk = blockIdx.x / nbn;           // K slice (0..SK-1)
  const int w = wave_id();
  const int l = lane_id();
  const int lm = l & 15;
  const int KS = (K + 31) / 32;
  const int KH = (KS + SK - 1) / SK;         // steps per slice
  const int ks0 = sk * KH;
  const int ks1 = min(ks0 + KH, KS);

  const bf16x8* pa = reinterpret_cast<const bf16x8*>(A_pack) + l;
  const bf16x8* pw = reinterpret_cast<const bf16x8*>(W_pack) +
                     (int64_t)(blockIdx.x % nbn) * KS * 64 + l;

  f32x4 acc0 = {}, acc1 = {};
  const int nown = (ks1 - ks0 - w + 3) / 4;  // owned steps: ks0 + w + 4i
  int i = 0;
  for (; i + 8 <= nown; i += 8) {
    bf16x8 a0v[8], a1v[8], bwv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int ks = ks0 + w + 4 * (i + u);
      a0v[u] = pa[ks * 128];
      a1v[u] = pa[ks * 128 + 64];
      bwv[u] = pw[ks * 64];
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      acc0 = mfma_16x16x32_bf16(a0v[u], bwv[u], acc0);
      acc1 = mfma_16x16x32_bf16(a1v[u], bwv[u], acc1);
    }
  }
  for (; i < nown; ++i) {
    const int ks = ks0 + w + 4 * i;
    bf16x8 a0v = pa[ks * 128];
    bf16x8 a1v = pa[ks * 128 + 64];
    bf16x8 bwv = pw[ks * 64];
    acc0 = mfma_16x16x32_bf16(a0v, bwv, acc0);
    acc1 = mfma_16x16x32_bf16(a1v, bwv, acc1);
  }

  const int fr0 = (l >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    red[(w * MAXB + fr0 + r) * 16 + lm] = acc0[r];
    red[(w * MAXB + 16 + fr0 + r) * 16 + lm] = acc1[r];
  }
  __syncthreads();
  float* out = (SK > 1 && sk == 1) ? C2 : C;
  for (int idx = threadIdx.x; idx < M * 16; idx += CELL_THREADS) {
    const int b = idx / 16, jj = idx % 16;
    if (n0 + jj >= N) continue;
    float v = red[(0 * MAXB + b) * 16 + jj] + red[(1 * MAXB + b) * 16 + jj] +
              red[(2 * MAXB + b) * 16 + jj] + red[(3 * MAXB + b) * 16 + jj];
    out[(int64_t)b * N + n0 + jj] = v;
  }
}

// ---------------------------------------------------------------------------
// Fused backward step: recurrent hop for step t + dgate elementwise for
// step t-1 in ONE launch (reference main.py:113 BPTT backward).
// ---------------------------------------------------------------------------
// The per-step pair (dgate kernel at the ~4.5 us dispatch floor + hop
// kernel) costs ~13 us/step-layer; this kernel halves the launch count.
// The fusion is column-aligned so NO freshly-produced data is broadcast
// inside the launch (the persistent-backward failure mode, PERF.md):
//   * phase 1 (hop): identical to smallm_packed_nt<32,2> — the block's
//     K-half MFMA over the PREVIOUS launch's packed dG[t] (crosses a
//     kernel boundary: bulk HBM-speed reads),
//   * the pair of blocks sharing an n-tile exchange their two K-half
//     partials through write-through f32 stores + one monotonic
//     per-tile arrival counter (no grid barrier, no L2 invalidate:
//     1.3 KB of fresh data read memory-side),
//   * phase 2 (dgate[t-1]): each block of the pair takes half the batch
//     rows of ITS OWN 16 columns — everything it needs (dh, rec, dY,
//     dc) is block-local — and writes dG[t-1] + the packed slot the
//     NEXT launch reads (double-buffered by step parity: this launch's
//     phase-1 readers must not see this launch's phase-2 writes).
typedef __attribute__((address_space(1))) unsigned int gau32;
#define ZRLX_AGENT __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT

// Optional phase census (tools/bwd_census.hip compiles this file with
// -DZAMD_BWD_PROF); compiled out of the production .so.
#ifdef ZAMD_BWD_PROF
__device__ unsigned long long g_bwd_prof[512 * 8];
// block 0's exit stamp from the previous launch: lets the census split
// the wall-minus-phases residue into (in-kernel preamble) vs (dispatch /
// launch-boundary gap) — the round-2 "2.7 us unattributed" lever.
__device__ unsigned long long g_bwd_prev_exit;
#define BPROF_STAMP(v) \
  unsigned long long v = \
      (threadIdx.x == 0) ? __builtin_amdgcn_s_memrealtime() : 0
#define BPROF_ACC(ph, t0, t1) \
  if (threadIdx.x == 0) g_bwd_prof[blockIdx.x * 8 + (ph)] += (t1) - (t0)
#define BPROF_LAUNCH_GAP(entry)                                       \
  if (blockIdx.x == 0 && threadIdx.x == 0) {                          \
    unsigned long long pe = g_bwd_prev_exit;                          \
    if (pe && (entry) > pe) g_bwd_prof[6] += (entry) - pe;            \
  }
#define BPROF_EXIT(v) \
  if (blockIdx.x == 0 && threadIdx.x == 0) g_bwd_prev_exit = (v)
#else
#define BPROF_STAMP(v) \
  do {                 \
  } while (0)
#define BPROF_ACC(ph, t0, t1) \
  do {                        \
  } while (0)
#define BPROF_LAUNCH_GAP(entry) \
  do {                          \
  } while (0)
#define BPROF_EXIT(v) \
  do {                \
  } while (0)
#endif

DEV_INLINE void store_wt_f32(float* p, float v) {
  __hip_atomic_store((gau32*)(uintptr_t)p,
                     __builtin_bit_cast(unsigned int, v), ZRLX_AGENT);
}
DEV_INLINE float load_wt_f32(const float* p) {
  unsigned int u =
      __hip_atomic_load((const gau32*)(uintptr_t)p, ZRLX_AGENT);
  return __builtin_bit_cast(float, u);
}

// NS = K-split ways (2 or 4): 4 puts ~2 blocks on each CU.
// NTHR = threads/block: 512 (8 waves) doubles the in-flight load
// streams per CU at the SAME grid — the measured hop bottleneck is
// per-wave outstanding-load capacity (4-way K-split halved the hop
// phase by doubling waves/CU, but paid for it in cross-block sync skew
// and dispatch; 8-wave blocks get the load parallelism without either).
template <int MAXB, int NS, int NTHR>
__global__ __launch_bounds__(NTHR) void smallm_fused_bwd_kernel(
    const bf16* __restrict__ A_pack,  // packed dG[t] (previous launch)
    const bf16* __restrict__ W_pack,  // packed W_h^T shadow
    float* __restrict__ P,            // [NS][M*N] f32 K-slice partials
    const bf16* __restrict__ dy,      // dY[t-1], [B,H]
    float* __restrict__ dc,           // [B,H] carried cell grad
    const bf16* __restrict__ rec,     // rec[t-1] block records
    bf16* __restrict__ dG,            // dG[t-1] out, [B,4H]
    bf16* __restrict__ dG_pack_out,   // packed slot for the NEXT launch
    unsigned int* __restrict__ flags, // [ceil(N/16)] monotonic counters
    unsigned int* __restrict__ abort_flag,
    int M, int N, int K, int HSp, unsigned int step) {
  constexpr int NW = NTHR / 64;  // waves per block
  __shared__ float red[NW * MAXB * 16];
  __shared__ int ok_s;

  BPROF_STAMP(bpe);  // first executed statement: kernel-entry stamp
  BPROF_LAUNCH_GAP(bpe);

  const int nbn = (N + 15) / 16;
  // Partner-interleaved block order: the NS blocks of an n-tile are
  // ADJACENT in blockIdx so the dispatcher starts them back-to-back.
  // (With sk-major order partners launched ~nbn dispatch slots apart,
  // and that start-time skew was the bulk of the 1.3 us pair-sync
  // phase: the early partner spins until the late one publishes.)
  const int nb = blockIdx.x / NS;
  const int n0 = nb * 16;
  const int sk = blockIdx.x % NS;   // K slice (0..NS-1)
  const int w = wave_id();
  const int l = lane_id();
  const int lm = l & 15;
  const int KS = (K + 31) / 32;
  const int KH = (KS + NS - 1) / NS;
  const int ks0 = sk * KH;
  const int ks1 = min(ks0 + KH, KS);

  const bf16x8* pa = reinterpret_cast<const bf16x8*>(A_pack) + l;
  const bf16x8* pw =
      reinterpret_cast<const bf16x8*>(W_pack) + (int64_t)nb * KS * 64 + l;

  BPROF_STAMP(bp0);
  f32x4 acc0 = {}, acc1 = {};
  const int nown = (ks1 - ks0 - w + NW - 1) / NW;
  int i = 0;
  for (; i + 8 <= nown; i += 8) {
    bf16x8 a0v[8], a1v[8], bwv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int ks = ks0 + w + NW * (i + u);
      a0v[u] = pa[ks * 128];
      a1v[u] = pa[ks * 128 + 64];
      bwv[u] = pw[ks * 64];
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      acc0 = mfma_16x16x32_bf16(a0v[u], bwv[u], acc0);
      acc1 = mfma_16x16x32_bf16(a1v[u], bwv[u], acc1);
    }
  }
  for (; i < nown; ++i) {
    const int ks = ks0 + w + NW * i;
    bf16x8 a0v = pa[ks * 128];
    bf16x8 a1v = pa[ks * 128 + 64];
    bf16x8 bwv = pw[ks * 64];
    acc0 = mfma_16x16x32_bf16(a0v, bwv, acc0);
    acc1 = mfma_16x16x32_bf16(a1v, bwv, acc1);
  }

  BPROF_STAMP(bp1);
  BPROF_ACC(5, bpe, bp0);  // in-kernel preamble (entry -> first phase)
  BPROF_ACC(0, bp0, bp1);  // hop MFMA (A/W loads + mfma)
  // Prefetch the dgate phase's partner-independent inputs (rec record,
  // dY, dc) NOW: the loads complete under the publish/arrive/spin that
  // follows instead of serializing into phase 2. nrows*16 <= 256, so
  // each thread owns at most one dgate element.
  const int Bh = (M + NS - 1) / NS;
  const int rb0 = sk * Bh;
  const int nrows = min(M - rb0, Bh);
  const int db = rb0 + threadIdx.x / 16;
  const int dj = n0 + threadIdx.x % 16;
  const bool dwork = ((int)threadIdx.x < nrows * 16) && (dj < N);
  float p_i = 0.f, p_f = 0.f, p_o = 0.f, p_n = 0.f, p_tc = 0.f,
        p_cprev = 0.f, p_dy = 0.f, p_dc = 0.f;
  if (dwork) {
    const int blk = dj / HSp, jr = dj % HSp;
    const bf16* r = rec + (((int64_t)blk * M + db) * 6) * HSp;
    p_i = bf2f(r[0 * HSp + jr]);
    p_f = bf2f(r[1 * HSp + jr]);
    p_o = bf2f(r[2 * HSp + jr]);
    p_n = bf2f(r[3 * HSp + jr]);
    p_tc = bf2f(r[4 * HSp + jr]);
    p_cprev = bf2f(r[5 * HSp + jr]);
    const int64_t e = (int64_t)db * N + dj;
    p_dy = bf2f(dy[e]);
    p_dc = dc[e];
  }

  const int fr0 = (l >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    red[(w * MAXB + fr0 + r) * 16 + lm] = acc0[r];
    red[(w * MAXB + 16 + fr0 + r) * 16 + lm] = acc1[r];
  }
  __syncthreads();

  BPROF_STAMP(bp2);
  BPROF_ACC(1, bp1, bp2);  // prefetch issue + red write + sync
  // publish this K-half's partial (write-through: memory-side visible
  // once the wave's vmcnt drains; the pair partner reads it sc1)
  float* mine = P + (int64_t)sk * M * N;
  for (int idx = threadIdx.x; idx < M * 16; idx += NTHR) {
    const int b = idx / 16, jj = idx % 16;
    if (n0 + jj >= N) continue;
    float v = 0.f;
#pragma unroll
    for (int ww = 0; ww < NW; ++ww) v += red[(ww * MAXB + b) * 16 + jj];
    store_wt_f32(mine + (int64_t)b * N + n0 + jj, v);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  BPROF_STAMP(bp3);
  BPROF_ACC(2, bp2, bp3);  // partial publish + drain
  if (threadIdx.x == 0) {
    ok_s = 1;
    gau32* f = (gau32*)(uintptr_t)(flags + nb);
    __hip_atomic_fetch_add(f, 1u, ZRLX_AGENT);
    unsigned int spins = 0;
    while (__hip_atomic_load(f, ZRLX_AGENT) < (unsigned int)NS * step) {
      __builtin_amdgcn_s_sleep(2);
      if (++spins > 20000000u) {
        atomicOr(abort_flag, 1u);
        ok_s = 0;
        break;
      }
    }
  }
  __syncthreads();
  BPROF_STAMP(bp4);
  BPROF_ACC(3, bp3, bp4);  // pair arrive + spin
  if (!ok_s) return;

  // phase 2: dgate[t-1] for this block's half of the batch rows of its
  // own 16 columns (mirrors lstm_cell_bwd_elt_kernel's math; inputs
  // preloaded above, only the partner partial is read here)
  if (dwork) {
    const int b = db, j = dj, jj = dj - n0;
    float vo = 0.f;
#pragma unroll
    for (int ww = 0; ww < NW; ++ww) vo += red[(ww * MAXB + b) * 16 + jj];
    const int64_t e = (int64_t)b * N + j;
    float ps = 0.f;  // partner partials; summed apart so the NS=2 total
                     // stays p_dy + (vo + partner), bitwise-identical to
                     // the per-step pair kernel's association
#pragma unroll
    for (int s = 1; s < NS; ++s)
      ps += load_wt_f32(P + (int64_t)((sk + s) % NS) * M * N + e);
    const float dh = p_dy + (vo + ps);
    const float do_ = dh * p_tc;
    const float dct = p_dc + dh * p_o * (1.f - p_tc * p_tc);
    const float di = dct * p_n;
    const float df = dct * p_cprev;
    const float dn = dct * p_i;
    const bf16 v[4] = {f2bf(di * p_i * (1.f - p_i)),
                       f2bf(df * p_f * (1.f - p_f)),
                       f2bf(do_ * p_o * (1.f - p_o)),
                       f2bf(dn * (1.f - p_n * p_n))};
    const int64_t gbase = (int64_t)b * 4 * N + j;
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const int k = g * N + j;
      dG[gbase + (int64_t)g * N] = v[g];
      const int ks = k / 32, sub = k % 32;
      const int pl = (b & 15) + 16 * (sub / 8);
      dG_pack_out[(((int64_t)ks * 2 + b / 16) * 64 + pl) * 8 + sub % 8] =
          v[g];
    }
    dc[e] = dct * p_f;
  }
  BPROF_STAMP(bp5);
  BPROF_ACC(4, bp4, bp5);  // dgate + dG/pack stores
  BPROF_EXIT(bp5);
}

// ---------------------------------------------------------------------------
// 2-step-batched fused backward (ZAREMBA_AMD_BWD_BATCH2=1): one launch
// runs hop[t]+dgate[t-1], a padded grid barrier + agent acquire, then
// hop[t-1]+dgate[t-2] — halving the launch count (the launch-boundary
// gap measures ~2 us/launch). The carried cell grad dc stays in a
// REGISTER between the two in-launch steps (the (row, col) partition is
// identical), and the mid-launch dgate pack is published WRITE-THROUGH
// so the second hop's plain loads (after the acquire) see it.
// ---------------------------------------------------------------------------
typedef __attribute__((address_space(1))) unsigned short gau16;
DEV_INLINE void store_wt_bf16(bf16* p, bf16 v) {
  __hip_atomic_store((gau16*)(uintptr_t)p,
                     __builtin_bit_cast(unsigned short, v), ZRLX_AGENT);
}

// padded two-level grid barrier — a copy of the proven forward barrier
// (lstm_persistent.hip xcd_grid_barrier): pstate words [0, 513),
// monotonic generations, every hot word on its own 128-B line.
DEV_INLINE bool fused2_grid_barrier(unsigned int* pstate, int NBLK,
                                    unsigned int gen,
                                    unsigned int* abort_flag) {
  __shared__ int ok_s2;
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    ok_s2 = 1;
    gau32* st = (gau32*)(uintptr_t)pstate;
    const int grp = blockIdx.x & 7;
    const int ngroups = NBLK < 8 ? NBLK : 8;
    const int nbg = (NBLK - grp + 7) / 8;
    unsigned int t0 = __hip_atomic_fetch_add(&st[grp * 32], 1u, ZRLX_AGENT);
    if (t0 == gen * (unsigned int)nbg - 1) {
      unsigned int tt = __hip_atomic_fetch_add(&st[256], 1u, ZRLX_AGENT);
      if (tt == gen * (unsigned int)ngroups - 1)
        for (int x = 0; x < 8; ++x)
          __hip_atomic_store(&st[288 + x * 32], gen, ZRLX_AGENT);
    }
    unsigned int spins = 0;
    while (__hip_atomic_load(&st[288 + grp * 32], ZRLX_AGENT) < gen) {
      __builtin_amdgcn_s_sleep(2);
      if (++spins > 20000000u) {
        atomicOr(abort_flag, 1u);
        ok_s2 = 0;
        break;
      }
    }
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
  return ok_s2 != 0;
}

template <int MAXB, int NTHR>
__global__ __launch_bounds__(NTHR) void smallm_fused_bwd2_kernel(
    const bf16* __restrict__ A_pack,  // prev launch's packed dgates
    const bf16* __restrict__ W_pack,
    float* __restrict__ P,            // [2][M*N] partials (reused per half)
    const bf16* __restrict__ dY,      // base [T, M, N]
    float* __restrict__ dc,           // [M, N] carried grad (launch hand-off)
    const bf16* __restrict__ rec,     // base [T] block records
    bf16* __restrict__ dG,            // base [T, M, K]
    bf16* __restrict__ pack_mid,      // dgate[t-1] pack (in-launch exchange)
    bf16* __restrict__ pack_out,      // dgate[t-2] pack (next launch reads)
    unsigned int* __restrict__ flags,
    unsigned int* __restrict__ pstate,
    unsigned int* __restrict__ abort_flag,
    int M, int N, int K, int HSp, int t, unsigned int step1,
    unsigned int gen, int64_t rstep) {
  constexpr int NW = NTHR / 64;
  __shared__ float red[NW * MAXB * 16];
  __shared__ int ok_s;

  const int nbn = (N + 15) / 16;
  const int nb = blockIdx.x / 2;  // partner-interleaved (see fused1)
  const int n0 = nb * 16;
  const int sk = blockIdx.x % 2;
  const int w = wave_id();
  const int l = lane_id();
  const int lm = l & 15;
  const int KS = (K + 31) / 32;
  const int KH = (KS + 1) / 2;
  const int ks0 = sk * KH;
  const int ks1 = min(ks0 + KH, KS);
  const int Bh = (M + 1) / 2;
  const int rb0 = sk * Bh;
  const int nrows = min(M - rb0, Bh);
  const int db = rb0 + threadIdx.x / 16;
  const int dj = n0 + threadIdx.x % 16;
  const bool dwork = ((int)threadIdx.x < nrows * 16) && (dj < N);
  const int64_t e = (int64_t)db * N + dj;
  const int fr0 = (l >> 4) * 4;
  const bf16x8* pw =
      reinterpret_cast<const bf16x8*>(W_pack) + (int64_t)nb * KS * 64 + l;

  const bf16* asrc[2] = {A_pack, pack_mid};
  bf16* pdst[2] = {pack_mid, pack_out};
  const int ts[2] = {t - 1, t - 2};
  float dc_carry = 0.f;

#pragma unroll
  for (int h2 = 0; h2 < 2; ++h2) {
    // ---- hop MFMA over asrc[h2] (same body as fused1) ------------------
    f32x4 acc0 = {}, acc1 = {};
    {
      const bf16x8* pa = reinterpret_cast<const bf16x8*>(asrc[h2]) + l;
      const int nown = (ks1 - ks0 - w + NW - 1) / NW;
      int i = 0;
      for (; i + 8 <= nown; i += 8) {
        bf16x8 a0v[8], a1v[8], bwv[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          const int ks = ks0 + w + NW * (i + u);
          a0v[u] = pa[ks * 128];
          a1v[u] = pa[ks * 128 + 64];
          bwv[u] = pw[ks * 64];
        }
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          acc0 = mfma_16x16x32_bf16(a0v[u], bwv[u], acc0);
          acc1 = mfma_16x16x32_bf16(a1v[u], bwv[u], acc1);
        }
      }
      for (; i < nown; ++i) {
        const int ks = ks0 + w + NW * i;
        bf16x8 a0v = pa[ks * 128];
        bf16x8 a1v = pa[ks * 128 + 64];
        bf16x8 bwv = pw[ks * 64];
        acc0 = mfma_16x16x32_bf16(a0v, bwv, acc0);
        acc1 = mfma_16x16x32_bf16(a1v, bwv, acc1);
      }
    }
    // ---- prefetch this half's dgate inputs -----------------------------
    float p_i = 0.f, p_f = 0.f, p_o = 0.f, p_n = 0.f, p_tc = 0.f,
          p_cprev = 0.f, p_dy = 0.f, p_dc = 0.f;
    if (dwork) {
      const int blk = dj / HSp, jr = dj % HSp;
      const bf16* r =
          rec + (int64_t)ts[h2] * rstep + (((int64_t)blk * M + db) * 6) * HSp;
      p_i = bf2f(r[0 * HSp + jr]);
      p_f = bf2f(r[1 * HSp + jr]);
      p_o = bf2f(r[2 * HSp + jr]);
      p_n = bf2f(r[3 * HSp + jr]);
      p_tc = bf2f(r[4 * HSp + jr]);
      p_cprev = bf2f(r[5 * HSp + jr]);
      p_dy = bf2f(dY[(int64_t)ts[h2] * M * N + e]);
      p_dc = (h2 == 0) ? dc[e] : dc_carry;
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      red[(w * MAXB + fr0 + r) * 16 + lm] = acc0[r];
      red[(w * MAXB + 16 + fr0 + r) * 16 + lm] = acc1[r];
    }
    __syncthreads();
    // ---- publish this K-half's partial (write-through) -----------------
    float* mine = P + (int64_t)sk * M * N;
    for (int idx = threadIdx.x; idx < M * 16; idx += NTHR) {
      const int b = idx / 16, jj = idx % 16;
      if (n0 + jj >= N) continue;
      float v = 0.f;
#pragma unroll
      for (int ww = 0; ww < NW; ++ww) v += red[(ww * MAXB + b) * 16 + jj];
      store_wt_f32(mine + (int64_t)b * N + n0 + jj, v);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    // ---- pair arrive + spin --------------------------------------------
    if (threadIdx.x == 0) {
      ok_s = 1;
      gau32* f = (gau32*)(uintptr_t)(flags + nb);
      __hip_atomic_fetch_add(f, 1u, ZRLX_AGENT);
      unsigned int spins = 0;
      const unsigned int want = 2u * (step1 + (unsigned int)h2);
      while (__hip_atomic_load(f, ZRLX_AGENT) < want) {
        __builtin_amdgcn_s_sleep(2);
        if (++spins > 20000000u) {
          atomicOr(abort_flag, 1u);
          ok_s = 0;
          break;
        }
      }
    }
    __syncthreads();
    if (!ok_s) return;
    // ---- dgate for this half's step ------------------------------------
    if (dwork) {
      const int b = db, j = dj, jj = dj - n0;
      float vo = 0.f;
#pragma unroll
      for (int ww = 0; ww < NW; ++ww) vo += red[(ww * MAXB + b) * 16 + jj];
      const float ps = load_wt_f32(P + (int64_t)(sk ^ 1) * M * N + e);
      const float dh = p_dy + (vo + ps);
      const float do_ = dh * p_tc;
      const float dct = p_dc + dh * p_o * (1.f - p_tc * p_tc);
      const float di = dct * p_n;
      const float df = dct * p_cprev;
      const float dn = dct * p_i;
      const bf16 v[4] = {f2bf(di * p_i * (1.f - p_i)),
                         f2bf(df * p_f * (1.f - p_f)),
                         f2bf(do_ * p_o * (1.f - p_o)),
                         f2bf(dn * (1.f - p_n * p_n))};
      bf16* dGt = dG + (int64_t)ts[h2] * M * K;
      bf16* pk = pdst[h2];
      const int64_t gbase = (int64_t)b * K + j;
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        const int k = g * N + j;
        dGt[gbase + (int64_t)g * N] = v[g];
        const int ks = k / 32, sub = k % 32;
        const int pl = (b & 15) + 16 * (sub / 8);
        bf16* pp = pk + (((int64_t)ks * 2 + b / 16) * 64 + pl) * 8 + sub % 8;
        if (h2 == 0)
          store_wt_bf16(pp, v[g]);  // in-launch consumer after the barrier
        else
          *pp = v[g];               // next-launch consumer (kernel boundary)
      }
      if (h2 == 0)
        dc_carry = dct * p_f;
      else
        dc[e] = dct * p_f;
    }
    // ---- between halves: all dgate[t-1] packs visible everywhere -------
    if (h2 == 0) {
      if (!fused2_grid_barrier(pstate, nbn * 2, gen, abort_flag)) return;
    }
  }
}

void launch_smallm_fused_bwd2(const bf16* A_pack, const bf16* W_pack,
                              float* P, const bf16* dY, float* dc,
                              const bf16* rec, bf16* dG, bf16* pack_mid,
                              bf16* pack_out, unsigned int* flags,
                              unsigned int* pstate,
                              unsigned int* abort_flag, int M, int N, int K,
                              int HSp, int t, unsigned int step1,
                              unsigned int gen, int64_t rstep, int nthreads,
                              hipStream_t stream) {
#define ZAMD_FB2_LAUNCH(NT_)                                                \
  hipLaunchKernelGGL((smallm_fused_bwd2_kernel<32, NT_>),                   \
                     dim3(cdiv(N, 16) * 2), dim3(NT_), 0, stream, A_pack,   \
                     W_pack, P, dY, dc, rec, dG, pack_mid, pack_out, flags, \
                     pstate, abort_flag, M, N, K, HSp, t, step1, gen, rstep)
  if (nthreads == 1024) ZAMD_FB2_LAUNCH(1024);
  else if (nthreads == 768) ZAMD_FB2_LAUNCH(768);
  else if (nthreads == 512) ZAMD_FB2_LAUNCH(512);
  else ZAMD_FB2_LAUNCH(256);
#undef ZAMD_FB2_LAUNCH
}

void launch_smallm_fused_bwd(const bf16* A_pack, const bf16* W_pack,
                             float* P, const bf16* dy, float* dc,
                             const bf16* rec, bf16* dG, bf16* dG_pack_out,
                             unsigned int* flags, unsigned int* abort_flag,
                             int M, int N, int K, int HSp, unsigned int step,
                             int nsplit, int nthreads, hipStream_t stream) {
#define ZAMD_FB_LAUNCH(NS_, NT_)                                           \
  hipLaunchKernelGGL((smallm_fused_bwd_kernel<32, NS_, NT_>),              \
                     dim3(cdiv(N, 16) * (NS_)), dim3(NT_), 0, stream,      \
                     A_pack, W_pack, P, dy, dc, rec, dG, dG_pack_out,      \
                     flags, abort_flag, M, N, K, HSp, step)
  if (nsplit == 4) {
    if (nthreads == 512) ZAMD_FB_LAUNCH(4, 512);
    else                 ZAMD_FB_LAUNCH(4, 256);
  } else {
    if (nthreads == 1024)     ZAMD_FB_LAUNCH(2, 1024);
    else if (nthreads == 768) ZAMD_FB_LAUNCH(2, 768);
    else if (nthreads == 512) ZAMD_FB_LAUNCH(2, 512);
    else                      ZAMD_FB_LAUNCH(2, 256);
  }
#undef ZAMD_FB_LAUNCH
}

void launch_smallm_packed_nt(const bf16* A_pack, const bf16* W_pack,
                             float* C, float* C2, int M, int N, int K,
                             hipStream_t stream) {
  if (C2) {
    hipLaunchKernelGGL((smallm_packed_nt_kernel<32, 2>),
                       dim3(cdiv(N, 16) * 2), dim3(CELL_THREADS), 0, stream,
                       A_pack, W_pack, C, C2, M, N, K);
  } else {
    hipLaunchKernelGGL((smallm_packed_nt_kernel<32, 1>), dim3(cdiv(N, 16)),
                       dim3(CELL_THREADS), 0, stream, A_pack, W_pack, C,
                       nullptr, M, N, K);
  }
}

}  // namespace zamd
