// Persistent LSTM kernels for gfx950 — one launch runs a whole layer
// unroll (forward or backward).
//
// Why (measured, see profiles/): with one launch per timestep, the W_h
// weight (18 MB bf16 at H=1500) misses L2 on EVERY launch — kernel
// boundaries do not retain it across the 8 per-XCD L2s — so each of the
// 70 cell launches of a training step re-pulls the full weight from
// LLC/HBM and runs ~90% wave-wait. Here each of the NB resident
// workgroups holds its weight slice in LDS for the entire unroll; the
// per-step global traffic is just the h (fwd) / dgates (bwd) broadcast.
//
// Inter-step exchange (guide §6 Guideline 16, placement-independent):
//   * the exchanged tensor's own [t] slot is the buffer — every slot is
//     written exactly once inside a launch, so no ring/parity logic and
//     no reader-side address reuse. (A granule-tagged variant measured
//     slower: relaxed agent loads are memory-side served, so every
//     consumer re-pulled the payload from the fabric.)
//   * producers store the exchanged values WRITE-THROUGH (relaxed
//     agent-scope = sc1) and every wave drains vmcnt before arriving at
//     an XCD-grouped two-level counter barrier (monotonic counters,
//     epoch generations); consumers take one agent acquire after the
//     generation flip, then read PLAIN (L2-amplified). Spins are
//     bounded; on timeout a block sets *abort and exits.
//   * barrier state (8 group counters, top counter, 8 generations) is
//     zeroed before every launch by the sequence driver.
//
// Scattered-store avoidance (measured: 2-byte stores into [B,4H]-layout
// gates cost ~90 MB/launch of read-modify-write line fills): everything
// the backward needs per (t, block) is written CONTIGUOUSLY as a
// 6-channel block record rec[t][blk][b][6][HS] = {i, f, o, n,
// tanh(c_t+1), c_t(bf16)}. c itself stays fp32 in registers across the
// unroll (exact state); only the backward's df term sees bf16 c_prev.
// Standard-layout c_all/h_all are written where the framework needs
// them (h every step — it IS the layer output and the exchange buffer —
// c only at the final step, the carried state).
#include "common.h"

namespace zamd {

typedef __attribute__((address_space(1))) unsigned int gu32;

constexpr int PCELL_THREADS = 256;
constexpr int REC_CH = 6;  // i,f,o,n,tanh_c_new,c_prev channels

// Forward threads/block: 512 (8 waves: memory phases get 2x the streams
// AND the gate MFMA splits across 2 waves per gate by K-parity;
// measured best) or 256 (4 waves, serial K per gate — the bitwise
// persistent-vs-per-step oracle path). env ZAREMBA_AMD_FWD_WAVES
// initializes; set_fwd_threads() overrides at runtime (tests).
int g_fwd_threads = [] {
  const char* e = getenv("ZAREMBA_AMD_FWD_WAVES");
  return e ? atoi(e) * 64 : 512;
}();
void set_fwd_threads_impl(int v) { g_fwd_threads = v; }

// HS must be even; NB = ceil(H/HS) <= 250 so the grid is co-resident.
// Prefer HS=8 (NB=188 at H=1500): the grid barrier is the forward's
// dominant per-step cost and scales with workgroup count (census probe:
// 9.9 us @250 WGs, 5.1 @125).
int persistent_hs(int H) {
  int hs = 2 * cdiv(H, 2 * 250);
  return hs < 8 ? 8 : hs;
}

// dynamic-LDS bytes the forward kernel needs (B <= 32)
size_t persistent_fwd_lds(int B, int H) {
  const int HS = persistent_hs(H);
  const int KPAD = ((H + 31) / 32) * 32 + 8;
  return (size_t)(4 * HS + B) * KPAD * 2 + (size_t)4 * B * 16 * 4 +
         (size_t)B * HS * 2 + 16;
}

#define RLX_AGENT __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT

// pstate layout — PADDED two-level counters, every hot word on its own
// 128-B line (zeroed per launch by the driver, words [0, 513)):
//   word 32*g  (g=0..7): group arrival counters (blockIdx & 7)
//   word 256:            top counter (one bump per group per step)
//   word 288+32*g:       per-group generation release words
// Monotonic generations, no resets. Census (tools/census.hip, 188 WGs):
// padded two-level 1.62 us vs flat single-counter 2.82 vs two-level on
// ADJACENT words 5.9 — the original two-level lost to flat only because
// its 8 group counters shared one cache line, serializing every add.
DEV_INLINE bool xcd_grid_barrier(unsigned int* pstate, int NB,
                                 unsigned int gen,
                                 unsigned int* abort_flag,
                                 bool acquire = true) {
  __shared__ int ok_s;
  // every wave drains its own write-through stores before arriving
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    ok_s = 1;
    gu32* st = (gu32*)(uintptr_t)pstate;
    const int grp = blockIdx.x & 7;
    const int ngroups = NB < 8 ? NB : 8;
    const int nbg = (NB - grp + 7) / 8;  // blocks with blockIdx%8 == grp
    unsigned int t = __hip_atomic_fetch_add(&st[grp * 32], 1u, RLX_AGENT);
    if (t == gen * (unsigned int)nbg - 1) {
      unsigned int tt = __hip_atomic_fetch_add(&st[256], 1u, RLX_AGENT);
      if (tt == gen * (unsigned int)ngroups - 1)
        for (int x = 0; x < 8; ++x)
          __hip_atomic_store(&st[288 + x * 32], gen, RLX_AGENT);
    }
    unsigned int spins = 0;
    while (__hip_atomic_load(&st[288 + grp * 32], RLX_AGENT) < gen) {
      __builtin_amdgcn_s_sleep(2);
      if (++spins > 20000000u) {
        atomicOr(abort_flag, 1u);
        ok_s = 0;
        break;
      }
    }
    if (acquire) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  }
  __syncthreads();
  return ok_s != 0;
}

// sc1 (relaxed agent-scope) 16-B load as two 8-B atomic loads: served
// memory-side, so the freshly write-through-published h is visible with
// NO acquire fence (and no per-step L2 invalidate). 8-B alignment holds
// because persistent_ok requires H even.
typedef __attribute__((address_space(1))) unsigned long long gu64;
DEV_INLINE bf16x8 load_sc1_vec8(const bf16* p) {
  const gu64* q = (const gu64*)(uintptr_t)p;
  unsigned long long lo = __hip_atomic_load(q, RLX_AGENT);
  unsigned long long hi = __hip_atomic_load(q + 1, RLX_AGENT);
  union {
    unsigned long long u[2];
    bf16x8 v;
  } r;
  r.u[0] = lo;
  r.u[1] = hi;
  return r.v;
}

// paired write-through store of two adjacent bf16 values
DEV_INLINE void store_pair_wt(bf16* p, bf16 lo, bf16 hi) {
  unsigned int packed = (unsigned int)__builtin_bit_cast(unsigned short, lo) |
                        ((unsigned int)__builtin_bit_cast(unsigned short, hi)
                         << 16);
  __hip_atomic_store((gu32*)(uintptr_t)p, packed, RLX_AGENT);
}

// Optional phase census (tools/fwd_census.hip compiles this file with
// -DZAMD_FWD_PROF): s_memrealtime (100 MHz) deltas per phase per block,
// accumulated into a device global. Compiled out of the production .so.
#ifdef ZAMD_FWD_PROF
__device__ unsigned long long g_fwd_prof[256 * 8];
#define PROF_STAMP(v) \
  unsigned long long v = \
      (threadIdx.x == 0) ? __builtin_amdgcn_s_memrealtime() : 0
#define PROF_ACC(ph, t0, t1) \
  if (threadIdx.x == 0) g_fwd_prof[blockIdx.x * 8 + (ph)] += (t1) - (t0)
#else
#define PROF_STAMP(v) \
  do {                \
  } while (0)
#define PROF_ACC(ph, t0, t1) \
  do {                       \
  } while (0)
#endif

// ===========================================================================
// Forward
// ===========================================================================
// NTHR: 256 = 4 waves (one wave per gate, serial K — the bitwise
// oracle path); 512 = 8 waves: the memory phases (h staging, publish,
// W preload) get 2x the streams AND each gate's K reduction is split
// across 2 waves by granule parity with a deterministic two-phase gbuf
// combine (same LDS layout either way).
template <int MAXB, int NTHR>
__global__ __launch_bounds__(NTHR) void lstm_persistent_fwd_kernel(
    const bf16* __restrict__ gx,     // [T, B, 4H]
    const bf16* __restrict__ W_h,    // [4H, H] bf16 shadow
    bf16* __restrict__ h_all,        // [T+1, B, H]; slot 0 = h0 (input)
    float* __restrict__ c_all,       // [T+1, B, H]; slot 0 = c0 (input);
                                     // only slot T is written back
    bf16* __restrict__ rec,          // [T][NB][B][6][HS] block records
    unsigned int* __restrict__ pstate,
    unsigned int* __restrict__ abort_flag,
    int T, int B, int H, int HS) {
  const int KS = (H + 31) / 32;
  const int KPAD = KS * 32 + 8;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* Ws = reinterpret_cast<bf16*>(smem);            // [4*HS][KPAD]
  bf16* hs = Ws + (int64_t)4 * HS * KPAD;              // [B][KPAD]
  float* gbuf = reinterpret_cast<float*>(hs + (int64_t)B * KPAD);  // [4][B][16]
  bf16* hbuf = reinterpret_cast<bf16*>(gbuf + 4 * B * 16);  // [B][HS]

  const int NB = (H + HS - 1) / HS;

  const int j0 = blockIdx.x * HS;
  // 8-wave blocks split each gate's K reduction across 2 waves by
  // granule PARITY (wave w -> gate w&3, K-parity w>>2): even granules
  // on waves 0-3, odd on 4-7, combined in a fixed two-phase gbuf
  // exchange (no atomics, no extra LDS — deterministic, but a
  // different f32 association than the serial 4-wave path, so the
  // bitwise persistent-vs-per-step oracle runs at 4 waves).
  constexpr int NWG = NTHR / 256;  // K-parity ways (1 or 2)
  const int g = wave_id() & 3;
  const int kh = wave_id() >> 2;
  const int l = lane_id();
  const int lm = l & 15;
  const int t_ = threadIdx.x;

  // ---- load the block's W_h slice into LDS (once) ------------------------
  for (int idx = t_; idx < 4 * HS * (KPAD / 8); idx += NTHR) {
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
  for (int idx = t_; idx < B * (KPAD - H); idx += NTHR) {
    const int b = idx / (KPAD - H);
    const int k = H + idx % (KPAD - H);
    hs[(int64_t)b * KPAD + k] = (bf16)0.f;
  }

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
#ifdef ZAMD_FWD_PROF
    PROF_STAMP(pt0);
#endif
    // Prefetch this thread's gx slice BEFORE the barrier: the loads are
    // independent of h_t, and gx[t] is LLC-cold scattered data whose
    // latency otherwise lands inside the pointwise phase.
    bf16 gxi = (bf16)0.f, gxf = (bf16)0.f, gxo = (bf16)0.f,
         gxn = (bf16)0.f;
    if (own) {
      const int64_t gxb =
          ((int64_t)t * B + own_b) * 4 * H + j0 + own_jj;
      gxi = gx[gxb + 0 * H];
      gxf = gx[gxb + 1 * H];
      gxo = gx[gxb + 2 * H];
      gxn = gx[gxb + 3 * H];
    }
#ifdef ZAMD_FWD_PROF
    PROF_STAMP(pt1);
    PROF_ACC(0, pt0, pt1);  // gx prefetch issue
#endif
    if (t > 0) {
      // no acquire: h is staged below with sc1 (memory-side) loads, and
      // nothing else read inside the loop is written cross-block
      if (!xcd_grid_barrier(pstate, NB, (unsigned int)t, abort_flag,
                            /*acquire=*/false))
        return;
    } else {
      __syncthreads();
    }
#ifdef ZAMD_FWD_PROF
    PROF_STAMP(pt2);
    PROF_ACC(1, pt1, pt2);  // barrier (arrival drain + spin + acquire)
#endif
    // ---- stage h_t into LDS, software-pipelined against the MFMA ---------
    // K is split at a granule boundary: half A is staged first (8/16-
    // vector register chunks: all loads of a chunk issue before any
    // ds_write consumes one — the old per-vector loop paid one vmcnt
    // wait per 16 B, 7.3 us/step-layer); half B's sc1 loads are ISSUED
    // into registers before the first sync, fly during the half-A MFMA
    // (global vmcnt vs LDS lgkmcnt — no forced wait), and are written +
    // consumed after it. Loads are unguarded vec8: the last vector of a
    // row over-reads <= 14 B into the next h_all row (in-bounds: the
    // staging reads slots 0..T-1 of the T+1-slot buffer), and the hs
    // columns >= H it fills multiply Ws pad zeros in the MFMA.
    const int vecs = (H + 7) / 8;
    const int ksA = (KS + 1) / 2;  // granules in half A
    const int vA = ksA * 4 < vecs ? ksA * 4 : vecs;  // vectors in half A
    const int vB = vecs - vA;
    const int totalB = B * vB;
    // 8-vector chunks per half (each half is ~half the old 16-chunk
    // payload, so the clamp-duplicate overhead stays the same). ALL
    // register-array loops are compile-time unrolled with CLAMPED
    // addresses: a runtime-bounded loop over a register array sends the
    // array to scratch (measured: stage-h 2.4 -> 30 us, the whole
    // kernel thrashing private memory).
    constexpr int DEPTH = 2048 / NTHR;  // 8 at 256 thr, 4 at 512
    bf16x8 vreg[DEPTH];
    {
      const bf16* hsrc = h_all + (int64_t)t * B * H;
      const int totalA = B * vA;
      for (int idx = t_; idx < totalA; idx += DEPTH * NTHR) {
        bf16x8 v[DEPTH];
        int id2 = idx;
#pragma unroll
        for (int u = 0; u < DEPTH; ++u, id2 += NTHR) {
          const int ic = id2 < totalA ? id2 : totalA - 1;
          const int b = ic / vA, k = (ic % vA) * 8;
          v[u] = load_sc1_vec8(hsrc + (int64_t)b * H + k);
        }
        id2 = idx;
#pragma unroll
        for (int u = 0; u < DEPTH; ++u, id2 += NTHR) {
          if (id2 < totalA) {
            const int b = id2 / vA, k = (id2 % vA) * 8;
            *reinterpret_cast<bf16x8*>(hs + (int64_t)b * KPAD + k) = v[u];
          }
        }
      }
      // issue half B's loads now; they complete under the half-A MFMA
      if (vB > 0) {
#pragma unroll
        for (int u = 0; u < DEPTH; ++u) {
          const int idx = t_ + u * NTHR;
          const int ic = idx < totalB ? idx : totalB - 1;
          const int b = ic / vB, k = (vA + ic % vB) * 8;
          vreg[u] = load_sc1_vec8(hsrc + (int64_t)b * H + k);
        }
      }
    }
    __syncthreads();  // half A visible in LDS
#ifdef ZAMD_FWD_PROF
    PROF_STAMP(pt3);
    PROF_ACC(2, pt2, pt3);  // stage half A + issue half B
#endif

    // ---- gate MFMA reduction (wave -> gate g, K-parity kh) ---------------
    f32x4 acc0 = {}, acc1 = {};
    const bf16* pw = Ws + (int64_t)(g * HS + wc) * KPAD;
    const bf16* pa0 = hs + (int64_t)a0r * KPAD;
    const bf16* pa1 = hs + (int64_t)a1r * KPAD;
    auto mfma_range = [&](int ks_lo, int ks_hi) {
      int ks = ks_lo + kh;
      for (; ks + 3 * NWG < ks_hi; ks += 4 * NWG) {
        bf16x8 a0v[4], a1v[4], bwv[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int k = (ks + u * NWG) * 32 + lk;
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
      for (; ks < ks_hi; ks += NWG) {
        const int k = ks * 32 + lk;
        bf16x8 a0v = *reinterpret_cast<const bf16x8*>(pa0 + k);
        bf16x8 a1v = *reinterpret_cast<const bf16x8*>(pa1 + k);
        bf16x8 bwv = *reinterpret_cast<const bf16x8*>(pw + k);
        acc0 = mfma_16x16x32_bf16(a0v, bwv, acc0);
        acc1 = mfma_16x16x32_bf16(a1v, bwv, acc1);
      }
    };
    mfma_range(0, ksA);
    // ---- land half B in LDS, then finish the reduction -------------------
    if (vB > 0) {
      {
        const bf16* hsrc = h_all + (int64_t)t * B * H;
#pragma unroll
        for (int u = 0; u < DEPTH; ++u) {
          const int idx = t_ + u * NTHR;
          if (idx < totalB) {
            const int b = idx / vB, k = (vA + idx % vB) * 8;
            *reinterpret_cast<bf16x8*>(hs + (int64_t)b * KPAD + k) = vreg[u];
          }
        }
        // overflow (totalB > DEPTH vectors/thread — only at B/H beyond
        // the Large config): stage the remainder load->write directly
        for (int idx = t_ + DEPTH * NTHR; idx < totalB;
             idx += NTHR) {
          const int b = idx / vB, k = (vA + idx % vB) * 8;
          *reinterpret_cast<bf16x8*>(hs + (int64_t)b * KPAD + k) =
              load_sc1_vec8(hsrc + (int64_t)b * H + k);
        }
      }
      __syncthreads();  // half B visible
      mfma_range(ksA, KS);
    }
    // two-phase deterministic combine: even-K waves write, odd-K add
    if (kh == 0) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        if (fr0 + r < B) gbuf[(g * B + fr0 + r) * 16 + lm] = acc0[r];
        if (16 + fr0 + r < B)
          gbuf[(g * B + 16 + fr0 + r) * 16 + lm] = acc1[r];
      }
    }
    if (NWG == 2) {
      __syncthreads();
      if (kh == 1) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          if (fr0 + r < B) gbuf[(g * B + fr0 + r) * 16 + lm] += acc0[r];
          if (16 + fr0 + r < B)
            gbuf[(g * B + 16 + fr0 + r) * 16 + lm] += acc1[r];
        }
      }
    }
    __syncthreads();
#ifdef ZAMD_FWD_PROF
    PROF_STAMP(pt4);
    PROF_ACC(3, pt3, pt4);  // gate MFMA + gbuf exchange
#endif

    // ---- pointwise cell update ------------------------------------------
    if (own) {
      const int b = own_b, jj = own_jj;
      const int j = j0 + jj;
      const int64_t gxbase = ((int64_t)t * B + b) * 4 * H + j;
      float gi = gbuf[(0 * B + b) * 16 + jj] + bf2f(gxi);
      float gf = gbuf[(1 * B + b) * 16 + jj] + bf2f(gxf);
      float go = gbuf[(2 * B + b) * 16 + jj] + bf2f(gxo);
      float gn = gbuf[(3 * B + b) * 16 + jj] + bf2f(gxn);
      float i_ = 1.f / (1.f + __expf(-gi));
      float f_ = 1.f / (1.f + __expf(-gf));
      float o_ = 1.f / (1.f + __expf(-go));
      float n_ = tanhf(gn);
      const float c_prev = c_reg;
      c_reg = f_ * c_reg + i_ * n_;
      const float tc = tanhf(c_reg);
      const float h_ = o_ * tc;
      // contiguous block record (i,f,o,n,tanh_c_new,c_prev)
      bf16* r = rec + ((((int64_t)t * NB + blockIdx.x) * B + b) * REC_CH) * HS;
      r[0 * HS + jj] = f2bf(i_);
      r[1 * HS + jj] = f2bf(f_);
      r[2 * HS + jj] = f2bf(o_);
      r[3 * HS + jj] = f2bf(n_);
      r[4 * HS + jj] = f2bf(tc);
      r[5 * HS + jj] = f2bf(c_prev);
      hbuf[b * HS + jj] = f2bf(h_);
      if (t == T - 1) c_all[((int64_t)T * B + b) * H + j] = c_reg;
    }
    __syncthreads();
#ifdef ZAMD_FWD_PROF
    PROF_STAMP(pt5);
    PROF_ACC(4, pt4, pt5);  // pointwise + rec stores
#endif
    // ---- publish h_{t+1}: paired write-through stores --------------------
    {
      bf16* hdst = h_all + (int64_t)(t + 1) * B * H;
      for (int i = t_; i < B * HS / 2; i += NTHR) {
        const int b = i / (HS / 2);
        const int jj = (i % (HS / 2)) * 2;
        const int j = j0 + jj;
        if (j < H) {
          store_pair_wt(hdst + (int64_t)b * H + j, hbuf[b * HS + jj],
                        (j + 1 < H) ? hbuf[b * HS + jj + 1] : (bf16)0.f);
        }
      }
    }
#ifdef ZAMD_FWD_PROF
    PROF_STAMP(pt6);
    PROF_ACC(5, pt5, pt6);  // h publish issue
#endif
  }
}

void launch_lstm_persistent_fwd(const bf16* gx, const bf16* W_h, bf16* h_all,
                                float* c_all, bf16* rec,
                                unsigned int* pstate,
                                unsigned int* abort_flag, int T, int B,
                                int H, hipStream_t stream) {
  const int HS = persistent_hs(H);
  const int NB = cdiv(H, HS);
  size_t lds = persistent_fwd_lds(B, H);
  const int nthr = g_fwd_threads;
  if (nthr == 512) {
    hipLaunchKernelGGL((lstm_persistent_fwd_kernel<32, 512>), dim3(NB),
                       dim3(512), lds, stream, gx, W_h, h_all, c_all,
                       rec, pstate, abort_flag, T, B, H, HS);
  } else {
    hipLaunchKernelGGL((lstm_persistent_fwd_kernel<32, 256>), dim3(NB),
                       dim3(256), lds, stream, gx, W_h, h_all, c_all,
                       rec, pstate, abort_flag, T, B, H, HS);
  }
}

// ===========================================================================
// Backward (v2)
// ===========================================================================
// One launch per layer reverse unroll. Per step t = T-1..0, each of the
// NB resident blocks (owning h-units [j0, j0+HS)):
//   1. dgate elementwise for its units from the 6-channel block record +
//      dY[t] + register-carried (dc, dh_rec) state (rec/dY prefetched
//      during the previous step's MFMA phase),
//   2. publishes its dgates write-through, BOTH as standard [T,B,4H] dG
//      (consumed later by the dW GEMMs) and in MFMA fragment-packed
//      layout (consumed in-launch: contiguous 1 KB wave loads straight
//      from L2 — a chunk-staged LDS variant with its 12 barriers/step
//      measured 2x slower),
//   3. grid barrier (same XCD-grouped counter barrier as forward),
//   4. recurrent hop dh_rec = dG @ W_h for its HS output columns: packed
//      global A-fragments x the LDS-resident W_h^T slice, 4 waves
//      round-robin the K steps, LDS reduction back to per-thread state.
// The per-step fallback pair re-reads the 18 MB W_h^T from HBM every
// step; here it is read once.
template <int MAXB>
__global__ __launch_bounds__(PCELL_THREADS) void lstm_persistent_bwd_kernel(
    const bf16* __restrict__ dY,     // [T, B, H]
    const bf16* __restrict__ rec,    // [T][NB][B][6][HS]
    const bf16* __restrict__ W_h_T,  // [H, 4H] transposed shadow
    bf16* __restrict__ dG,           // [T, B, 4H] out (write-through)
    bf16* __restrict__ dG_packT,     // [T][KS2*2*64*8] zero-prefilled
    unsigned int* __restrict__ pstate,
    unsigned int* __restrict__ abort_flag,
    int T, int B, int H, int HS) {
  const int K = 4 * H;
  const int KS2 = (K + 31) / 32;
  const int KWPAD = KS2 * 32 + 8;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* Ws = reinterpret_cast<bf16*>(smem);                 // [HS][KWPAD]
  float* gbuf = reinterpret_cast<float*>(Ws + (int64_t)HS * KWPAD);
  bf16* dgbuf = reinterpret_cast<bf16*>(gbuf + 4 * B * 16);  // [B][4][HS]

  const int NB = (H + HS - 1) / HS;

  const int j0 = blockIdx.x * HS;
  const int w = wave_id();
  const int l = lane_id();
  const int lm = l & 15;
  const int t_ = threadIdx.x;

  // ---- load the W_h^T slice (rows j0..j0+HS of [H, 4H]) into LDS ---------
  for (int idx = t_; idx < HS * (KWPAD / 8); idx += PCELL_THREADS) {
    const int kv = (idx % (KWPAD / 8)) * 8;
    const int row = idx / (KWPAD / 8);
    bf16x8 v = {};
    const int col = j0 + row;
    if (col < H) {
      const bf16* p = W_h_T + (int64_t)col * K + kv;
      if (kv + 8 <= K) {
        v = *reinterpret_cast<const bf16x8*>(p);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = (kv + e < K) ? p[e] : (bf16)0.f;
      }
    }
    *reinterpret_cast<bf16x8*>(Ws + (int64_t)row * KWPAD + kv) = v;
  }

  const int own_b = t_ / HS;
  const int own_jj = t_ % HS;
  const bool own = (t_ < B * HS) && (j0 + own_jj < H);
  float dc_reg = 0.f, dh_rec = 0.f;

  const int a0r = lm < B ? lm : B - 1;
  const int a1r = (16 + lm) < B ? (16 + lm) : B - 1;
  const int wc = (lm < HS ? lm : HS - 1);
  const int fr0 = (l >> 4) * 4;
  const int lk = (l >> 4) * 8;
  (void)a0r; (void)a1r; (void)lk;

  // prefetch the first step's record/dY for the owning thread
  float r_i = 0, r_f = 0, r_o = 0, r_n = 0, r_tc = 0, r_cp = 0, r_dy = 0;
  if (own) {
    const bf16* r =
        rec + ((((int64_t)(T - 1) * NB + blockIdx.x) * B + own_b) * REC_CH) *
                  HS;
    r_i = bf2f(r[0 * HS + own_jj]);
    r_f = bf2f(r[1 * HS + own_jj]);
    r_o = bf2f(r[2 * HS + own_jj]);
    r_n = bf2f(r[3 * HS + own_jj]);
    r_tc = bf2f(r[4 * HS + own_jj]);
    r_cp = bf2f(r[5 * HS + own_jj]);
    r_dy = bf2f(dY[((int64_t)(T - 1) * B + own_b) * H + j0 + own_jj]);
  }

  for (int t = T - 1; t >= 0; --t) {
    __syncthreads();  // previous step's dgbuf consumers done
    // ---- 1. dgate elementwise for own units ------------------------------
    if (own) {
      const float dh = r_dy + dh_rec;
      const float do_ = dh * r_tc;
      const float dct = dc_reg + dh * r_o * (1.f - r_tc * r_tc);
      dgbuf[(own_b * 4 + 0) * HS + own_jj] =
          f2bf(dct * r_n * r_i * (1.f - r_i));
      dgbuf[(own_b * 4 + 1) * HS + own_jj] =
          f2bf(dct * r_cp * r_f * (1.f - r_f));
      dgbuf[(own_b * 4 + 2) * HS + own_jj] = f2bf(do_ * r_o * (1.f - r_o));
      dgbuf[(own_b * 4 + 3) * HS + own_jj] =
          f2bf(dct * r_i * (1.f - r_n * r_n));
      dc_reg = dct * r_f;
    }
    __syncthreads();
    // ---- 2. publish dG[t]: standard + fragment-packed (write-through) ----
    {
      bf16* dst = dG + (int64_t)t * B * K;
      bf16* pkt = dG_packT + (int64_t)t * KS2 * 2 * 64 * 8;
      for (int i = t_; i < B * 4 * HS / 2; i += PCELL_THREADS) {
        const int b = i / (4 * HS / 2);
        const int rem = i % (4 * HS / 2);
        const int gg = rem / (HS / 2);
        const int jj = (rem % (HS / 2)) * 2;
        const int j = j0 + jj;
        if (j < H) {
          const bf16 v0 = dgbuf[(b * 4 + gg) * HS + jj];
          const bf16 v1 = (j + 1 < H) ? dgbuf[(b * 4 + gg) * HS + jj + 1]
                                      : (bf16)0.f;
          const int k = gg * H + j;
          store_pair_wt(dst + (int64_t)b * K + k, v0, v1);
          const int ks = k / 32, sub = k % 32;
          const int pl = (b & 15) + 16 * (sub / 8);
          store_pair_wt(
              pkt + ((((int64_t)ks * 2 + b / 16) * 64 + pl) * 8 + sub % 8),
              v0, v1);
        }
      }
    }
    // ---- 3. prefetch next step's record/dY (overlaps barrier + MFMA) -----
    if (own && t > 0) {
      const bf16* r =
          rec +
          ((((int64_t)(t - 1) * NB + blockIdx.x) * B + own_b) * REC_CH) * HS;
      r_i = bf2f(r[0 * HS + own_jj]);
      r_f = bf2f(r[1 * HS + own_jj]);
      r_o = bf2f(r[2 * HS + own_jj]);
      r_n = bf2f(r[3 * HS + own_jj]);
      r_tc = bf2f(r[4 * HS + own_jj]);
      r_cp = bf2f(r[5 * HS + own_jj]);
      r_dy = bf2f(dY[((int64_t)(t - 1) * B + own_b) * H + j0 + own_jj]);
    }
    // ---- 4. grid barrier --------------------------------------------------
    if (!xcd_grid_barrier(pstate, NB, (unsigned int)(T - t), abort_flag))
      return;
    // ---- 5. recurrent hop: packed global A x LDS W -----------------------
    f32x4 acc0 = {}, acc1 = {};
    {
      const bf16x8* pa = reinterpret_cast<const bf16x8*>(
                             dG_packT + (int64_t)t * KS2 * 2 * 64 * 8) +
                         l;
      const bf16* pw = Ws + (int64_t)wc * KWPAD;
      const int nown = (KS2 - w + 3) / 4;  // owned steps: w + 4i
      int i = 0;
      for (; i + 8 <= nown; i += 8) {
        bf16x8 a0v[8], a1v[8], bwv[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          const int ks = w + 4 * (i + u);
          a0v[u] = pa[ks * 128];
          a1v[u] = pa[ks * 128 + 64];
          bwv[u] = *reinterpret_cast<const bf16x8*>(pw + ks * 32 +
                                                    ((l >> 4) * 8));
        }
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          acc0 = mfma_16x16x32_bf16(a0v[u], bwv[u], acc0);
          acc1 = mfma_16x16x32_bf16(a1v[u], bwv[u], acc1);
        }
      }
      for (; i < nown; ++i) {
        const int ks = w + 4 * i;
        bf16x8 a0v = pa[ks * 128];
        bf16x8 a1v = pa[ks * 128 + 64];
        bf16x8 bwv = *reinterpret_cast<const bf16x8*>(pw + ks * 32 +
                                                      ((l >> 4) * 8));
        acc0 = mfma_16x16x32_bf16(a0v, bwv, acc0);
        acc1 = mfma_16x16x32_bf16(a1v, bwv, acc1);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      if (fr0 + r < B) gbuf[(w * B + fr0 + r) * 16 + lm] = acc0[r];
      if (16 + fr0 + r < B) gbuf[(w * B + 16 + fr0 + r) * 16 + lm] = acc1[r];
    }
    __syncthreads();
    if (own) {
      dh_rec = gbuf[(0 * B + own_b) * 16 + own_jj] +
               gbuf[(1 * B + own_b) * 16 + own_jj] +
               gbuf[(2 * B + own_b) * 16 + own_jj] +
               gbuf[(3 * B + own_b) * 16 + own_jj];
    }
  }
}

size_t persistent_bwd_lds(int B, int H) {
  const int HS = persistent_hs(H);
  const int KWPAD = ((4 * H + 31) / 32) * 32 + 8;
  return (size_t)HS * KWPAD * 2 + (size_t)4 * B * 16 * 4 +
         (size_t)B * 4 * HS * 2 + 16;
}

void launch_lstm_persistent_bwd(const bf16* dY, const bf16* rec,
                                const bf16* W_h_T, bf16* dG, bf16* dG_packT,
                                unsigned int* pstate,
                                unsigned int* abort_flag, int T, int B,
                                int H, hipStream_t stream) {
  const int HS = persistent_hs(H);
  const int NB = cdiv(H, HS);
  hipLaunchKernelGGL((lstm_persistent_bwd_kernel<32>), dim3(NB),
                     dim3(PCELL_THREADS), persistent_bwd_lds(B, H), stream,
                     dY, rec, W_h_T, dG, dG_packT, pstate, abort_flag, T, B,
                     H, HS);
}

}  // namespace zamd
