// Persistent LSTM forward kernel for gfx950 — one launch runs the whole
// T-step unroll of a layer.
//
// Why (measured, see profiles/): with one launch per timestep, the W_h
// weight (18 MB bf16 at H=1500) misses L2 on EVERY launch — kernel
// boundaries do not retain it across the 8 per-XCD L2s — so each of the
// 70 cell launches of a training step re-pulls the full weight from
// LLC/HBM and runs ~90% wave-wait. Here each of the NB resident
// workgroups holds its W_h slice in LDS for the entire unroll and the
// only per-step global traffic is the h broadcast (60 KB, L2-amplified).
//
// Inter-step exchange (guide §6 Guideline 16, placement-independent):
//   * h_all[t+1] itself is the exchange buffer — every slot is written
//     exactly once inside the launch, so there is no reader-side reuse
//     of addresses and no ring/parity logic. (A first version broadcast
//     h as 8-byte tagged granules; relaxed agent loads are memory-side
//     served, so every consumer re-pulled the full payload from the
//     fabric — 30 MB/step — and it measured slower than relaunching.)
//   * between steps: every block does {plain h stores -> __syncthreads ->
//     lane-0 agent release fence + vmcnt drain -> arrival} into an
//     XCD-grouped two-level counter barrier (monotonic counters, epoch
//     generations — no per-step state reset), then one agent acquire.
//     Spins are bounded; on timeout the block sets *abort and exits.
//   * the barrier state words (8 group counters, 1 top counter, 8
//     generation words) must be zeroed before every launch (the driver
//     issues the hipMemsetAsync).
//
// Compute per step is the same 4-wave MFMA gate reduction + pointwise
// cell update as the per-step cell kernel (bitwise-identical results).
#include "common.h"

namespace zamd {

typedef __attribute__((address_space(1))) unsigned int gu32;

constexpr int PCELL_THREADS = 256;

// HS must be even; NB = ceil(H/HS) <= 250 so the grid is co-resident.
int persistent_hs(int H) {
  return 2 * cdiv(H, 2 * 250);
}

#define RLX_AGENT __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT

// pstate layout: [0..7] group arrival counters, [8] top counter,
// [9..16] group generation words. Monotonic: gen = step index (1-based).
DEV_INLINE bool xcd_grid_barrier(unsigned int* pstate, int grp, int nbg,
                                 int ngroups, unsigned int gen,
                                 unsigned int* abort_flag) {
  __shared__ int ok_s;
  // every wave drains its own write-through h stores before arriving
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    ok_s = 1;
    gu32* st = (gu32*)(uintptr_t)pstate;
    unsigned int t = __hip_atomic_fetch_add(&st[grp], 1u, RLX_AGENT);
    if (t == gen * nbg - 1) {  // group leader (arrivals are monotonic)
      unsigned int tt = __hip_atomic_fetch_add(&st[8], 1u, RLX_AGENT);
      if (tt == gen * ngroups - 1) {  // last group: flip every generation
        for (int x = 0; x < 8; ++x)
          __hip_atomic_store(&st[9 + x], gen, RLX_AGENT);
      }
    }
    unsigned int spins = 0;
    while (__hip_atomic_load(&st[9 + grp], RLX_AGENT) < gen) {
      __builtin_amdgcn_s_sleep(8);
      if (++spins > 50000000u) {
        atomicOr(abort_flag, 1u);
        ok_s = 0;
        break;
      }
    }
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
  return ok_s != 0;
}

template <int MAXB>
__global__ __launch_bounds__(PCELL_THREADS) void lstm_persistent_fwd_kernel(
    const bf16* __restrict__ gx,     // [T, B, 4H]
    const bf16* __restrict__ W_h,    // [4H, H] bf16 shadow
    bf16* __restrict__ h_all,        // [T+1, B, H]; slot 0 = h0 (input)
    float* __restrict__ c_all,       // [T+1, B, H]; slot 0 = c0 (input)
    bf16* __restrict__ gates_out,    // [T, B, 4H]
    unsigned int* __restrict__ pstate,  // 17 zeroed words (barrier state)
    unsigned int* __restrict__ abort_flag,
    int T, int B, int H, int HS) {
  const int KS = (H + 31) / 32;
  const int KPAD = KS * 32 + 8;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* Ws = reinterpret_cast<bf16*>(smem);            // [4*HS][KPAD]
  bf16* hs = Ws + (int64_t)4 * HS * KPAD;              // [B][KPAD]
  float* gbuf = reinterpret_cast<float*>(hs + (int64_t)B * KPAD);
  bf16* hbuf = reinterpret_cast<bf16*>(gbuf + 4 * MAXB * 16);  // [B][HS]

  const int NB = (H + HS - 1) / HS;
  const int grp = blockIdx.x & 7;
  const int ngroups = NB < 8 ? NB : 8;
  const int nbg = (NB - grp + 7) / 8;  // blocks in this group

  const int j0 = blockIdx.x * HS;
  const int g = wave_id();
  const int l = lane_id();
  const int lm = l & 15;
  const int t_ = threadIdx.x;

  // ---- load the block's W_h slice into LDS (once) ------------------------
  for (int idx = t_; idx < 4 * HS * (KPAD / 8); idx += PCELL_THREADS) {
    const int kv = (idx % (KPAD / 8)) * 8;
    const int row = idx / (KPAD / 8);       // gg*HS + c
    const int gg = row / HS, c = row % HS;
    bf16x8 v = {};
    const int col = j0 + c;
    if (col < H) {
      const bf16* p = W_h + ((int64_t)gg * H + col) * H + kv;
      if (kv + 8 <= H) {
        v = *reinterpret_cast<const bf16x8*>(p);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = (kv + e < H) ? p[e] : (bf16)0.f;
      }
    }
    *reinterpret_cast<bf16x8*>(Ws + (int64_t)row * KPAD + kv) = v;
  }
  // zero the h image K-tail once (never rewritten)
  for (int idx = t_; idx < B * (KPAD - H); idx += PCELL_THREADS) {
    const int b = idx / (KPAD - H);
    const int k = H + idx % (KPAD - H);
    hs[(int64_t)b * KPAD + k] = (bf16)0.f;
  }

  // ---- per-thread cell state (thread t_ owns element (b, jj)) ------------
  const int own_b = t_ / HS;
  const int own_jj = t_ % HS;
  const bool own = (t_ < B * HS) && (j0 + own_jj < H);
  float c_reg = 0.f;
  if (own) c_reg = c_all[(int64_t)own_b * H + j0 + own_jj];

  const int a0r = lm < B ? lm : B - 1;
  const int a1r = (16 + lm) < B ? (16 + lm) : B - 1;
  const int wc = (lm < HS ? lm : HS - 1);
  const int fr0 = (l >> 4) * 4;
  const int lk = (l >> 4) * 8;

  for (int t = 0; t < T; ++t) {
    if (t > 0) {
      // all blocks' h_all[t] stores done + visible before anyone reads
      if (!xcd_grid_barrier(pstate, grp, nbg, ngroups, (unsigned int)t,
                            abort_flag))
        return;
    } else {
      __syncthreads();
    }
    // ---- stage h_t into the LDS image (plain global loads) ---------------
    {
      const bf16* hsrc = h_all + (int64_t)t * B * H;
      const int vecs = (H + 7) / 8;
      for (int idx = t_; idx < B * vecs; idx += PCELL_THREADS) {
        const int b = idx / vecs;
        const int k = (idx % vecs) * 8;
        bf16x8 v = {};
        const bf16* p = hsrc + (int64_t)b * H + k;
        if (k + 8 <= H) {
          v = *reinterpret_cast<const bf16x8*>(p);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) v[e] = (k + e < H) ? p[e] : (bf16)0.f;
        }
        *reinterpret_cast<bf16x8*>(hs + (int64_t)b * KPAD + k) = v;
      }
    }
    __syncthreads();

    // ---- gate MFMA reduction (wave g -> gate g, [32 x 16(HS)] tile) ------
    f32x4 acc0 = {}, acc1 = {};
    {
      const bf16* pw = Ws + (int64_t)(g * HS + wc) * KPAD;
      const bf16* pa0 = hs + (int64_t)a0r * KPAD;
      const bf16* pa1 = hs + (int64_t)a1r * KPAD;
      int ks = 0;
      for (; ks + 4 <= KS; ks += 4) {
        bf16x8 a0v[4], a1v[4], bwv[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int k = (ks + u) * 32 + lk;
          a0v[u] = *reinterpret_cast<const bf16x8*>(pa0 + k);
          a1v[u] = *reinterpret_cast<const bf16x8*>(pa1 + k);
          bwv[u] = *reinterpret_cast<const bf16x8*>(pw + k);
        }
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          acc0 = mfma_16x16x32_bf16(a0v[u], bwv[u], acc0);
          acc1 = mfma_16x16x32_bf16(a1v[u], bwv[u], acc1);
        }
      }
      for (; ks < KS; ++ks) {
        const int k = ks * 32 + lk;
        bf16x8 a0v = *reinterpret_cast<const bf16x8*>(pa0 + k);
        bf16x8 a1v = *reinterpret_cast<const bf16x8*>(pa1 + k);
        bf16x8 bwv = *reinterpret_cast<const bf16x8*>(pw + k);
        acc0 = mfma_16x16x32_bf16(a0v, bwv, acc0);
        acc1 = mfma_16x16x32_bf16(a1v, bwv, acc1);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      gbuf[(g * MAXB + fr0 + r) * 16 + lm] = acc0[r];
      gbuf[(g * MAXB + 16 + fr0 + r) * 16 + lm] = acc1[r];
    }
    __syncthreads();

    // ---- pointwise cell update + stores ----------------------------------
    if (own) {
      const int b = own_b, jj = own_jj;
      const int j = j0 + jj;
      const int64_t gxbase = ((int64_t)t * B + b) * 4 * H + j;
      float gi = gbuf[(0 * MAXB + b) * 16 + jj] + bf2f(gx[gxbase + 0 * H]);
      float gf = gbuf[(1 * MAXB + b) * 16 + jj] + bf2f(gx[gxbase + 1 * H]);
      float go = gbuf[(2 * MAXB + b) * 16 + jj] + bf2f(gx[gxbase + 2 * H]);
      float gn = gbuf[(3 * MAXB + b) * 16 + jj] + bf2f(gx[gxbase + 3 * H]);
      float i_ = 1.f / (1.f + __expf(-gi));
      float f_ = 1.f / (1.f + __expf(-gf));
      float o_ = 1.f / (1.f + __expf(-go));
      float n_ = tanhf(gn);
      c_reg = f_ * c_reg + i_ * n_;
      const float h_ = o_ * tanhf(c_reg);
      const int64_t hoff = ((int64_t)(t + 1) * B + b) * H + j;
      c_all[hoff] = c_reg;
      gates_out[gxbase + 0 * H] = f2bf(i_);
      gates_out[gxbase + 1 * H] = f2bf(f_);
      gates_out[gxbase + 2 * H] = f2bf(o_);
      gates_out[gxbase + 3 * H] = f2bf(n_);
      hbuf[b * HS + jj] = f2bf(h_);
    }
    __syncthreads();
    // publish h_{t+1} as paired write-through (sc1) stores — the only
    // data other blocks read inside this launch. No release fence needed
    // (R1 write-through form); each wave drains vmcnt at the barrier.
    {
      bf16* hdst = h_all + (int64_t)(t + 1) * B * H;
      for (int i = t_; i < B * HS / 2; i += PCELL_THREADS) {
        const int b = i / (HS / 2);
        const int jj = (i % (HS / 2)) * 2;
        const int j = j0 + jj;
        if (j < H) {
          const bf16 h0v = hbuf[b * HS + jj];
          const bf16 h1v = (j + 1 < H) ? hbuf[b * HS + jj + 1] : (bf16)0.f;
          unsigned int packed =
              (unsigned int)__builtin_bit_cast(unsigned short, h0v) |
              ((unsigned int)__builtin_bit_cast(unsigned short, h1v) << 16);
          gu32* p = (gu32*)(uintptr_t)(hdst + (int64_t)b * H + j);
          __hip_atomic_store(p, packed, RLX_AGENT);
        }
      }
    }
  }
}

void launch_lstm_persistent_fwd(const bf16* gx, const bf16* W_h, bf16* h_all,
                                float* c_all, bf16* gates_out,
                                unsigned int* pstate,
                                unsigned int* abort_flag, int T, int B,
                                int H, hipStream_t stream) {
  const int HS = persistent_hs(H);
  const int NB = cdiv(H, HS);
  const int KS = (H + 31) / 32;
  const int KPAD = KS * 32 + 8;
  size_t lds = (size_t)(4 * HS + B) * KPAD * 2 + 4 * 32 * 16 * sizeof(float) +
               (size_t)B * HS * 2 + 16;
  hipLaunchKernelGGL((lstm_persistent_fwd_kernel<32>), dim3(NB),
                     dim3(PCELL_THREADS), lds, stream, gx, W_h, h_all, c_all,
                     gates_out, pstate, abort_flag, T, B, H, HS);
}

}  // namespace zamd
