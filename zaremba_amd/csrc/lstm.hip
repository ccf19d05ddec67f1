// Fused LSTM cell kernels for gfx950 — the sequential hot path
// (SURVEY.md §2.3 K3+K4, reference model.py:34-45 math; gate order i,f,o,n).
//
// Forward, one timestep (launched T times per layer by the C++ sequence
// driver in ext.cpp, hipGraph-captured):
//   gates = h_prev @ W_h^T + gx_t          (gx already holds x@W_x^T+b_x+b_h)
//   i,f,o = sigmoid(g0,g1,g2); n = tanh(g3)
//   c = f*c_prev + i*n ; h = o*tanh(c)
//
// Geometry (skinny-M design, B<=32): grid = ceil(H/16) workgroups, each
// owning 16 hidden units; 4 waves per workgroup, wave g computes gate g's
// [32(M) x 16] tile with v_mfma_f32_16x16x32_bf16 over the full K=H
// reduction. h_prev is staged once into LDS ([32][Hpad] bf16); W_h B-
// fragments are read straight from global (row-major [4H, H], K-contig —
// W_h is L2/LLC-resident across the T-step unroll, which is the point of
// the per-step relaunch design on 8 XCDs). The cell pointwise update +
// state write happen in the same kernel via an LDS gate exchange.
//
// c is carried in fp32 across the whole epoch (truncated-BPTT state,
// reference main.py:110-111); h is bf16 (it feeds GEMMs).
//
// Backward, one timestep, two kernels (sequence driver loops t=T-1..0):
//   lstm_cell_bwd_elt: dgates_t from (dy_t + dh_rec, dc), updates dc.
//   smallm_gemm_nt:    dh_rec = dgates_t @ W_h  (via the W_h^T shadow,
//                      K-contiguous; 4 waves split K, LDS reduce).
#include "common.h"

namespace zamd {

constexpr int CELL_THREADS = 256;

// ---------------------------------------------------------------------------
// Forward cell
// ---------------------------------------------------------------------------
// LDS budget (H=1500): h tile 32 * (ceil(H/32)*32 + 8) * 2B = 96.8 KB,
// plus the 4x32x16 fp32 gate-exchange buffer (8 KB) -> one block per CU.
template <int MAXB>  // padded batch rows (32)
__global__ __launch_bounds__(CELL_THREADS, 1) void lstm_cell_fwd_kernel(
    const bf16* __restrict__ h_prev,   // [B, H]
    const float* __restrict__ c_prev,  // [B, H]
    const bf16* __restrict__ gx,       // [B, 4H] this timestep's input gates
    const bf16* __restrict__ W_h,      // [4H, H] row-major
    bf16* __restrict__ h_out,          // [B, H]
    float* __restrict__ c_out,         // [B, H]
    bf16* __restrict__ gates_out,      // [B, 4H] post-activation i,f,o,n
    int B, int H) {
  const int KSTEPS = (H + 31) / 32;
  const int HPAD = KSTEPS * 32 + 8;  // +8 bf16 row pad: conflict-free b128 reads
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* hs = reinterpret_cast<bf16*>(smem);                  // [MAXB][HPAD]
  float* gbuf = reinterpret_cast<float*>(smem + MAXB * HPAD * 2);  // [4][MAXB][16]

  const int j0 = blockIdx.x * 16;  // hidden-unit slice
  const int g = wave_id();         // gate index (i,f,o,n)
  const int l = lane_id();
  const int lm = l & 15;
  const int lk = (l >> 4) * 8;

  // ---- stage h_prev -> LDS (guarded vec8 loads, zero fill tails) ----------
  {
    const int vec_per_row = HPAD / 8;  // HPAD % 8 == 0
    for (int idx = threadIdx.x; idx < MAXB * vec_per_row;
         idx += CELL_THREADS) {
      int b = idx / vec_per_row;
      int k = (idx % vec_per_row) * 8;
      bf16x8 v = {};
      if (b < B && k < H) {
        const bf16* p = h_prev + (int64_t)b * H + k;
        if (k + 8 <= H) {
          v = *reinterpret_cast<const bf16x8*>(p);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) v[e] = (k + e < H) ? p[e] : (bf16)0.f;
        }
      }
      *reinterpret_cast<bf16x8*>(hs + (int64_t)b * HPAD + k) = v;
    }
  }
  __syncthreads();

  // ---- per-gate MFMA reduction over K = H --------------------------------
  f32x4 acc[MAXB / 16] = {};
  const int wrow = g * H + j0 + lm;        // W_h row for this lane's column
  const bool col_ok = (j0 + lm) < H;
  const bf16* wp = W_h + (int64_t)wrow * H;
  for (int ks = 0; ks < KSTEPS; ++ks) {
    int k = ks * 32 + lk;
    bf16x8 bfrag = {};
    if (col_ok) {
      if (k + 8 <= H) {
        bfrag = *reinterpret_cast<const bf16x8*>(wp + k);
      } else if (k < H) {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          bfrag[e] = (k + e < H) ? wp[k + e] : (bf16)0.f;
      }
    }
#pragma unroll
    for (int mf = 0; mf < MAXB / 16; ++mf) {
      bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
          hs + (int64_t)(mf * 16 + lm) * HPAD + ks * 32 + lk);
      acc[mf] = mfma_16x16x32_bf16(afrag, bfrag, acc[mf]);
    }
  }

  // ---- exchange gate tiles through LDS -----------------------------------
  // C/D map: col = l&15, row = (l>>4)*4 + r.
  const int fr0 = (l >> 4) * 4;
#pragma unroll
  for (int mf = 0; mf < MAXB / 16; ++mf) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = mf * 16 + fr0 + r;
      gbuf[(g * MAXB + row) * 16 + lm] = acc[mf][r];
    }
  }
  __syncthreads();

  // ---- pointwise cell update ---------------------------------------------
  for (int idx = threadIdx.x; idx < B * 16; idx += CELL_THREADS) {
    int b = idx / 16;
    int jj = idx % 16;
    int j = j0 + jj;
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
  const int KSTEPS = (H + 31) / 32;
  const int HPAD = KSTEPS * 32 + 8;
  size_t lds = (size_t)32 * HPAD * 2 + 4 * 32 * 16 * sizeof(float);
  int grid = cdiv(H, 16);
  hipLaunchKernelGGL((lstm_cell_fwd_kernel<32>), dim3(grid),
                     dim3(CELL_THREADS), lds, stream, h_prev, c_prev, gx, W_h,
                     h_out, c_out, gates_out, B, H);
}

// ---------------------------------------------------------------------------
// Backward: per-timestep elementwise dgate kernel
// ---------------------------------------------------------------------------
// dh_t = dy_t + dh_rec (recurrent grad from step t+1; fp32 buffer)
// tc = tanh(c_t);  do = dh*tc;  dct = dc + dh*o*(1-tc^2)
// di = dct*n; df = dct*c_prev; dn = dct*i; do as above
// pre-activation: dgi = di*i*(1-i); dgf = df*f*(1-f); dgo = do*o*(1-o);
//                 dgn = dn*(1-n^2)
// carried: dc <- dct * f
__global__ void lstm_cell_bwd_elt_kernel(
    const bf16* __restrict__ dy,       // [B,H] upstream at step t (bf16)
    const float* __restrict__ dh_rec,  // [B,H] recurrent grad (nullptr at t=T-1)
    float* __restrict__ dc,            // [B,H] carried, updated in place
    const bf16* __restrict__ gates,    // [B,4H] saved i,f,o,n
    const float* __restrict__ c_prev,  // [B,H] c_{t-1}
    const float* __restrict__ c_new,   // [B,H] c_t
    bf16* __restrict__ dG,             // [B,4H] out: pre-activation grads
    int B, int H) {
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
// Used for the recurrent backward hop dh_rec = dG_t @ W_h (B = the W_h^T
// shadow, [H, 4H] row-major = N x K with K contiguous). Grid = ceil(N/16);
// 4 waves split K and reduce through LDS. A is staged in K-chunks.
constexpr int SMK_CHUNK = 1024;  // K elements per staged A chunk

template <int MAXB>
__global__ __launch_bounds__(CELL_THREADS, 1) void smallm_gemm_nt_kernel(
    const bf16* __restrict__ A,  // [M, K]
    const bf16* __restrict__ B_, // [N, K]
    float* __restrict__ C,       // [M, N]
    int M, int N, int K) {
  constexpr int CP = SMK_CHUNK + 8;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* as = reinterpret_cast<bf16*>(smem);                    // [MAXB][CP]
  float* red = reinterpret_cast<float*>(smem + MAXB * CP * 2); // [4][MAXB][16]

  const int n0 = blockIdx.x * 16;
  const int w = wave_id();
  const int l = lane_id();
  const int lm = l & 15;
  const int lk = (l >> 4) * 8;
  const bool col_ok = (n0 + lm) < N;
  const bf16* bp = B_ + (int64_t)(n0 + lm) * K;

  f32x4 acc[MAXB / 16] = {};
  for (int k0 = 0; k0 < K; k0 += SMK_CHUNK) {
    const int klen = min(SMK_CHUNK, K - k0);
    // stage A chunk
    {
      const int vec = CP / 8;
      for (int idx = threadIdx.x; idx < MAXB * vec; idx += CELL_THREADS) {
        int b = idx / vec;
        int k = (idx % vec) * 8;
        bf16x8 v = {};
        if (b < M && k < klen) {
          const bf16* p = A + (int64_t)b * K + k0 + k;
          if (k + 8 <= klen) {
            v = *reinterpret_cast<const bf16x8*>(p);
          } else {
#pragma unroll
            for (int e = 0; e < 8; ++e)
              v[e] = (k + e < klen) ? p[e] : (bf16)0.f;
          }
        }
        *reinterpret_cast<bf16x8*>(as + (int64_t)b * CP + k) = v;
      }
    }
    __syncthreads();
    // each wave reduces its quarter of the chunk
    const int kq = SMK_CHUNK / 4;  // 256
    const int kw0 = w * kq;
    for (int ks = 0; ks < kq; ks += 32) {
      int kc = kw0 + ks;          // within chunk
      if (kc >= klen) break;
      int kg = k0 + kc + lk;      // global k for the B fragment
      bf16x8 bfrag = {};
      if (col_ok && kg < K) {
        if (kg + 8 <= K) {
          bfrag = *reinterpret_cast<const bf16x8*>(bp + kg);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            bfrag[e] = (kg + e < K) ? bp[kg + e] : (bf16)0.f;
        }
      }
#pragma unroll
      for (int mf = 0; mf < MAXB / 16; ++mf) {
        bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
            as + (int64_t)(mf * 16 + lm) * CP + kc + lk);
        acc[mf] = mfma_16x16x32_bf16(afrag, bfrag, acc[mf]);
      }
    }
    __syncthreads();
  }

  // reduce the 4 wave partials via LDS
  const int fr0 = (l >> 4) * 4;
#pragma unroll
  for (int mf = 0; mf < MAXB / 16; ++mf)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      red[(w * MAXB + mf * 16 + fr0 + r) * 16 + lm] = acc[mf][r];
  __syncthreads();
  for (int idx = threadIdx.x; idx < M * 16; idx += CELL_THREADS) {
    int b = idx / 16, jj = idx % 16;
    if (n0 + jj >= N) continue;
    float v = 0.f;
#pragma unroll
    for (int ww = 0; ww < 4; ++ww) v += red[(ww * MAXB + b) * 16 + jj];
    C[(int64_t)b * N + n0 + jj] = v;
  }
}

void launch_smallm_gemm_nt(const bf16* A, const bf16* B, float* C, int M,
                           int N, int K, hipStream_t stream) {
  size_t lds = (size_t)32 * (SMK_CHUNK + 8) * 2 + 4 * 32 * 16 * sizeof(float);
  hipLaunchKernelGGL((smallm_gemm_nt_kernel<32>), dim3(cdiv(N, 16)),
                     dim3(CELL_THREADS), lds, stream, A, B, C, M, N, K);
}

}  // namespace zamd
