#include "hip/hip_runtime.h"
// Fused LSTM cell kernels for gfx950 — the sequential hot path
// (SURVEY.md §2.3 K3+K4, reference model.py:34-45 math; gate order i,f,o,n).
//
// Forward, one timestep (launched T times per layer by the C++ sequence
// driver in ext_bind.hip, hipGraph-captured):
//   gates = h_prev @ W_h^T + gx_t          (gx already holds x@W_x^T+b_x+b_h)
//   i,f,o = sigmoid(g0,g1,g2); n = tanh(g3)
//   c = f*c_prev + i*n ; h = o*tanh(c)
//
// Design (skinny-M, B<=32, measured on MI355X):
//   * grid = ceil(H/16) workgroups x 4 waves; wave g owns gate g's
//     [32 x 16] tile and reduces the full K=H with
//     v_mfma_f32_16x16x32_bf16, fp32 accumulation.
//   * No LDS staging: h_prev (60 KB) and the W_h slice are L2/LLC-
//     resident across the T-step unroll (the point of the per-step
//     relaunch design), so both MFMA operands are loaded straight from
//     global with 16-B fragment reads. Out-of-range rows are CLAMPED to
//     a valid row (finite garbage that only lands in discarded output
//     rows/cols); only the K-tail step is element-guarded, because a
//     zero A-fragment against uninitialized W bytes could make 0*NaN.
//   * Depth-4 software prefetch on the fragment loads — per-element
//     bounds branches or unprefetched chains leave the wave latency-
//     bound at ~900 cycles/step (measured 26 us/step before; see
//     profiles/).
//   * The cell pointwise update + state write happen in-kernel through
//     an 8 KB LDS gate exchange. c carries in fp32 across the epoch
//     (truncated-BPTT state, reference main.py:110-111); h is bf16.
//
// Backward, one timestep, two kernels (driver loops t=T-1..0):
//   lstm_cell_bwd_elt: dgates_t from (dy_t + dh_rec, dc), updates dc.
//   smallm_gemm_nt:    dh_rec = dgates_t @ W_h  (via the W_h^T shadow;
//                      4 waves round-robin the K steps, LDS reduce).
#include "common.h"

namespace zamd {

constexpr int CELL_THREADS = 256;

// Guarded fragment load for the K-tail: elements past `limit` are zero.
DEV_INLINE bf16x8 frag_tail(const bf16* p, int k, int limit) {
  bf16x8 v = {};
#pragma unroll
  for (int e = 0; e < 8; ++e)
    if (k + e < limit) v[e] = p[k + e];
  return v;
}

// ---------------------------------------------------------------------------
// Forward cell
// ---------------------------------------------------------------------------
template <int MAXB>
__global__ __launch_bounds__(CELL_THREADS) void lstm_cell_fwd_kernel(
    const bf16* __restrict__ h_prev,   // [B, H]
    const float* __restrict__ c_prev,  // [B, H]
    const bf16* __restrict__ gx,       // [B, 4H] this timestep's input gates
    const bf16* __restrict__ W_h,      // [4H, H] row-major
    bf16* __restrict__ h_out,          // [B, H]
    float* __restrict__ c_out,         // [B, H]
    bf16* __restrict__ gates_out,      // [B, 4H] post-activation i,f,o,n
    int B, int H) {
  __shared__ float gbuf[4 * MAXB * 16];

  const int j0 = blockIdx.x * 16;  // hidden-unit slice
  const int g = wave_id();         // gate index (i,f,o,n)
  const int l = lane_id();
  const int lm = l & 15;
  const int lk = (l >> 4) * 8;

  // Row-clamped operand base pointers (clamp => finite garbage that only
  // reaches discarded outputs).
  const int a0r = lm < B ? lm : B - 1;
  const int a1r = (16 + lm) < B ? (16 + lm) : B - 1;
  const int wr = g * H + (j0 + lm < H ? j0 + lm : H - 1);
  const bf16* pa0 = h_prev + (int64_t)a0r * H;
  const bf16* pa1 = h_prev + (int64_t)a1r * H;
  const bf16* pw = W_h + (int64_t)wr * H;

  const int full = H / 32;         // unguarded 32-k steps
  const bool tail = (full * 32) < H;

  f32x4 acc0 = {}, acc1 = {};

  // 8-step chunks: 24 x 16-B loads issued per iteration before any wait,
  // amortizing L2 latency over 16 MFMAs (hipcc groups loads at the top of
  // the iteration; cross-iteration register rotation does not survive the
  // scheduler, so work WITH that shape instead).
  int ks = 0;
  for (; ks + 8 <= full; ks += 8) {
    bf16x8 a0v[8], a1v[8], bwv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      a0v[u] = *reinterpret_cast<const bf16x8*>(pa0 + (ks + u) * 32 + lk);
      a1v[u] = *reinterpret_cast<const bf16x8*>(pa1 + (ks + u) * 32 + lk);
      bwv[u] = *reinterpret_cast<const bf16x8*>(pw + (ks + u) * 32 + lk);
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      acc0 = mfma_16x16x32_bf16(a0v[u], bwv[u], acc0);
      acc1 = mfma_16x16x32_bf16(a1v[u], bwv[u], acc1);
    }
  }
  for (; ks < full; ++ks) {
    bf16x8 a0v = *reinterpret_cast<const bf16x8*>(pa0 + ks * 32 + lk);
    bf16x8 a1v = *reinterpret_cast<const bf16x8*>(pa1 + ks * 32 + lk);
    bf16x8 bwv = *reinterpret_cast<const bf16x8*>(pw + ks * 32 + lk);
    acc0 = mfma_16x16x32_bf16(a0v, bwv, acc0);
    acc1 = mfma_16x16x32_bf16(a1v, bwv, acc1);
  }
  if (tail) {
    const int k = full * 32 + lk;
    bf16x8 a0t = frag_tail(pa0, k, H);
    bf16x8 a1t = frag_tail(pa1, k, H);
    bf16x8 bwt = frag_tail(pw, k, H);
    acc0 = mfma_16x16x32_bf16(a0t, bwt, acc0);
    acc1 = mfma_16x16x32_bf16(a1t, bwt, acc1);
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

  // ---- pointwise cell update ---------------------------------------------
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
    float c_ = f_ * c_prev[(int64_t)b * H + j] + i_ * n_;
    float h_ = o_ * tanhf(c_);
    c_out[(int64_t)b * H + j] = c_;
    h_out[(int64_t)b * H + j] = f2bf(h_);
    gates_out[gbase + 0 * H] = f2bf(i_);
    gates_out[gbase + 1 * H] = f2bf(f_);
    gates_out[gbase + 2 * H] = f2bf(o_);
    gates_out[gbase + 3 * H] = f2bf(n_);
  }
}

void launch_lstm_cell_fwd(const bf16* h_prev, const float* c_prev,
                          const bf16* gx, const bf16* W_h, bf16* h_out,
                          float* c_out, bf16* gates_out, int B, int H,
                          hipStream_t stream) {
  hipLaunchKernelGGL((lstm_cell_fwd_kernel<32>), dim3(cdiv(H, 16)),
                     dim3(CELL_THREADS), 0, stream, h_prev, c_prev, gx, W_h,
                     h_out, c_out, gates_out, B, H);
}

// ---------------------------------------------------------------------------
// Backward: per-timestep elementwise dgate kernel
// ---------------------------------------------------------------------------
// dh_t = dy_t + dh_rec; tc = tanh(c_t); do = dh*tc
// dct = dc + dh*o*(1-tc^2); di = dct*n; df = dct*c_prev; dn = dct*i
// pre-activation grads via the sigmoid/tanh local derivatives; carried
// dc <- dct * f.
__global__ void lstm_cell_bwd_elt_kernel(
    const bf16* __restrict__ dy, const float* __restrict__ dh_rec,
    float* __restrict__ dc, const bf16* __restrict__ gates,
    const float* __restrict__ c_prev, const float* __restrict__ c_new,
    bf16* __restrict__ dG, int B, int H) {
  int idx = blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= B * H) return;
  int b = idx / H, j = idx % H;
  const int64_t gbase = (int64_t)b * 4 * H + j;
  float i_ = bf2f(gates[gbase + 0 * H]);
  float f_ = bf2f(gates[gbase + 1 * H]);
  float o_ = bf2f(gates[gbase + 2 * H]);
  float n_ = bf2f(gates[gbase + 3 * H]);
  float dh = bf2f(dy[idx]) + (dh_rec ? dh_rec[idx] : 0.f);
  float tc = tanhf(c_new[idx]);
  float do_ = dh * tc;
  float dct = dc[idx] + dh * o_ * (1.f - tc * tc);
  float di = dct * n_;
  float df = dct * c_prev[idx];
  float dn = dct * i_;
  dG[gbase + 0 * H] = f2bf(di * i_ * (1.f - i_));
  dG[gbase + 1 * H] = f2bf(df * f_ * (1.f - f_));
  dG[gbase + 2 * H] = f2bf(do_ * o_ * (1.f - o_));
  dG[gbase + 3 * H] = f2bf(dn * (1.f - n_ * n_));
  dc[idx] = dct * f_;
}

void launch_lstm_cell_bwd_elt(const bf16* dy, const float* dh_rec, float* dc,
                              const bf16* gates, const float* c_prev,
                              const float* c_new, bf16* dG, int B, int H,
                              hipStream_t stream) {
  int n = B * H;
  hipLaunchKernelGGL(lstm_cell_bwd_elt_kernel, dim3(cdiv(n, 256)), dim3(256),
                     0, stream, dy, dh_rec, dc, gates, c_prev, c_new, dG, B,
                     H);
}

// ---------------------------------------------------------------------------
// Skinny-M NT GEMM: C[M<=32, N] (fp32) = A[M,K] bf16 @ B[N,K]^T bf16
// ---------------------------------------------------------------------------
// The recurrent backward hop dh_rec = dG_t @ W_h (B = the W_h^T shadow,
// [H, 4H] row-major, K contiguous). Grid = ceil(N/16); the 4 waves
// round-robin the 32-wide K steps (wave w takes steps w, w+4, ...) so
// every wave sees uniform full steps; one LDS reduction at the end.
// Same streaming design as the forward cell: row-clamped direct global
// fragment loads, depth-4 prefetch, element guards only on the K tail.
// Requires K % 8 == 0 (K = 4H here).
template <int MAXB>
__global__ __launch_bounds__(CELL_THREADS) void smallm_gemm_nt_kernel(
    const bf16* __restrict__ A,  // [M, K]
    const bf16* __restrict__ B_, // [N, K]
    float* __restrict__ C,       // [M, N]
    int M, int N, int K) {
  __shared__ float red[4 * MAXB * 16];

  const int n0 = blockIdx.x * 16;
  const int w = wave_id();
  const int l = lane_id();
  const int lm = l & 15;
  const int lk = (l >> 4) * 8;

  const int a0r = lm < M ? lm : M - 1;
  const int a1r = (16 + lm) < M ? (16 + lm) : M - 1;
  const int br = n0 + lm < N ? n0 + lm : N - 1;
  const bf16* pa0 = A + (int64_t)a0r * K;
  const bf16* pa1 = A + (int64_t)a1r * K;
  const bf16* pb = B_ + (int64_t)br * K;

  const int nsteps = (K + 31) / 32;
  const int full = K / 32;

  f32x4 acc0 = {}, acc1 = {};

  // wave w owns steps w, w+4, w+8, ... ; i-th owned step = w + 4i.
  // 8-owned-step chunks (24 loads in flight per iteration).
  const int nown = (full - w + 3) / 4;  // owned FULL steps
  int i = 0;
  for (; i + 8 <= nown; i += 8) {
    bf16x8 a0v[8], a1v[8], bwv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int kk = (w + 4 * (i + u)) * 32 + lk;
      a0v[u] = *reinterpret_cast<const bf16x8*>(pa0 + kk);
      a1v[u] = *reinterpret_cast<const bf16x8*>(pa1 + kk);
      bwv[u] = *reinterpret_cast<const bf16x8*>(pb + kk);
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      acc0 = mfma_16x16x32_bf16(a0v[u], bwv[u], acc0);
      acc1 = mfma_16x16x32_bf16(a1v[u], bwv[u], acc1);
    }
  }
  for (; i < nown; ++i) {
    const int kk = (w + 4 * i) * 32 + lk;
    bf16x8 a0v = *reinterpret_cast<const bf16x8*>(pa0 + kk);
    bf16x8 a1v = *reinterpret_cast<const bf16x8*>(pa1 + kk);
    bf16x8 bwv = *reinterpret_cast<const bf16x8*>(pb + kk);
    acc0 = mfma_16x16x32_bf16(a0v, bwv, acc0);
    acc1 = mfma_16x16x32_bf16(a1v, bwv, acc1);
  }
  // K tail step (K % 32 != 0), owned by wave (full % 4)
  if (full < nsteps && w == (full % 4)) {
    const int k = full * 32 + lk;
    bf16x8 a0t = frag_tail(pa0, k, K);
    bf16x8 a1t = frag_tail(pa1, k, K);
    bf16x8 bt = frag_tail(pb, k, K);
    acc0 = mfma_16x16x32_bf16(a0t, bt, acc0);
    acc1 = mfma_16x16x32_bf16(a1t, bt, acc1);
  }

  const int fr0 = (l >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    red[(w * MAXB + fr0 + r) * 16 + lm] = acc0[r];
    red[(w * MAXB + 16 + fr0 + r) * 16 + lm] = acc1[r];
  }
  __syncthreads();
  for (int idx = threadIdx.x; idx < M * 16; idx += CELL_THREADS) {
    const int b = idx / 16, jj = idx % 16;
    if (n0 + jj >= N) continue;
    float v = red[(0 * MAXB + b) * 16 + jj] + red[(1 * MAXB + b) * 16 + jj] +
              red[(2 * MAXB + b) * 16 + jj] + red[(3 * MAXB + b) * 16 + jj];
    C[(int64_t)b * N + n0 + jj] = v;
  }
}

void launch_smallm_gemm_nt(const bf16* A, const bf16* B, float* C, int M,
                           int N, int K, hipStream_t stream) {
  hipLaunchKernelGGL((smallm_gemm_nt_kernel<32>), dim3(cdiv(N, 16)),
                     dim3(CELL_THREADS), 0, stream, A, B, C, M, N, K);
}

}  // namespace zamd
