// Python bindings + host-side sequence drivers for the zaremba_amd HIP
// kernel library (gfx950).
//
// The per-timestep LSTM loops are driven from C++: the forward's
// persistent kernel and the backward's fused hop+dgate step kernels
// launch eagerly (cross-block-spin kernels hang under hipGraph replay
// on ROCm 7.x); the per-step FALLBACK trains (fused cell x T / dgate +
// recurrent GEMM x T) are cached as hipGraphs keyed on buffer pointers
// so a step is one replay instead of 35-70 host launches (SURVEY.md
// §3.5: the launch-overhead hot spot).
#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <array>
#include <cstdint>
#include <map>
#include <stdexcept>
#include <vector>

#include "kernels.h"

namespace zamd {

#define HIP_CHECK(expr)                                              \
  do {                                                               \
    hipError_t _e = (expr);                                          \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); \
  } while (0)

static hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

static const bf16* bf_ptr(const torch::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, "expected bf16 tensor");
  TORCH_CHECK(t.is_contiguous(), "expected contiguous tensor");
  return reinterpret_cast<const bf16*>(t.data_ptr());
}
static bf16* bf_ptr_mut(torch::Tensor& t) {
  return const_cast<bf16*>(bf_ptr(t));
}
static const float* f_ptr(const torch::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, "expected f32 tensor");
  TORCH_CHECK(t.is_contiguous(), "expected contiguous tensor");
  return t.data_ptr<float>();
}
static float* f_ptr_mut(torch::Tensor& t) { return const_cast<float*>(f_ptr(t)); }

// ---------------------------------------------------------------------------
// GEMM
// ---------------------------------------------------------------------------
// C[M,N] = A' @ B' + bias. trans_a: A is [K,M] (weight-grad path);
// trans_b: B is [K,N]; otherwise B is the NT weight layout [N,K].
//
// k_pad (NT only): run the reduction over k_pad >= K columns so the
// kernel has no K-tail phase (a tail tile measured +13.5 us; see
// PERF.md). Contract: B is the pre-padded [N, k_pad] tensor with ZERO
// in columns [K, k_pad) — A's over-read garbage multiplies those zeros
// — and A's storage extends >= 128 finite (zeroed) bytes past its end,
// because rows over-read into the next row and the last (and clamped)
// rows over-read into that slack. HipModel's slack-provisioned buffers
// guarantee both; callers must not pass k_pad for arbitrary tensors.
static void gemm(const torch::Tensor& A, const torch::Tensor& B,
                 torch::Tensor& C, const c10::optional<torch::Tensor>& bias,
                 bool trans_a, bool trans_b, int k_pad = 0) {
  int M = C.size(0), N = C.size(1);
  int K = trans_a ? A.size(0) : A.size(1);
  TORCH_CHECK((trans_a ? A.size(1) : A.size(0)) == M, "gemm: A/C M mismatch");
  TORCH_CHECK((trans_b ? B.size(1) : B.size(0)) == N, "gemm: B/C N mismatch");
  if (k_pad) {
    TORCH_CHECK(!trans_a && !trans_b, "gemm: k_pad is NT-only");
    TORCH_CHECK(k_pad >= K && k_pad % 64 == 0, "gemm: bad k_pad");
    TORCH_CHECK(B.size(1) == k_pad, "gemm: B not padded to k_pad");
  } else {
    TORCH_CHECK((trans_b ? B.size(0) : B.size(1)) == K, "gemm: A/B K mismatch");
  }
  int lda = A.size(1), ldb = B.size(1), ldc = C.size(1);
  if (k_pad) K = k_pad;  // lda stays the REAL row stride of A
  const float* bp = bias ? f_ptr(*bias) : nullptr;
  auto stream = current_stream();
  TORCH_CHECK(trans_a == trans_b, "gemm: only NT and TN layouts are wired");
  if (C.scalar_type() == torch::kFloat32) {
    if (trans_a)
      launch_gemm_t<true, true, float>(bf_ptr(A), bf_ptr(B), C.data_ptr<float>(),
                                       bp, M, N, K, lda, ldb, ldc, stream);
    else
      launch_gemm_t<false, false, float>(bf_ptr(A), bf_ptr(B),
                                         C.data_ptr<float>(), bp, M, N, K,
                                         lda, ldb, ldc, stream);
  } else {
    if (trans_a)
      launch_gemm_t<true, true, bf16>(bf_ptr(A), bf_ptr(B), bf_ptr_mut(C), bp,
                                      M, N, K, lda, ldb, ldc, stream);
    else
      launch_gemm_t<false, false, bf16>(bf_ptr(A), bf_ptr(B), bf_ptr_mut(C),
                                        bp, M, N, K, lda, ldb, ldc, stream);
  }
}

// 2-way split-K NT GEMM: f32 partials into C and C2 (the exact MFMA
// accumulators of the two K halves), combined by add2_f32_bf16. Same
// k_pad contract as gemm(); the split keys off the padded K so both
// halves are whole-BK-tile ranges.
// Returns the split ways nz used; C2 must hold >= nz-1 stacked [M, N]
// partial slices (the caller sums C + C2[0..nz-2]).
static int gemm_splitk(const torch::Tensor& A, const torch::Tensor& B,
                       torch::Tensor& C, torch::Tensor& C2,
                       const c10::optional<torch::Tensor>& bias, int k_pad) {
  int M = C.size(0), N = C.size(1);
  int K = A.size(1);
  TORCH_CHECK(A.size(0) == M, "gemm_splitk: A/C M mismatch");
  TORCH_CHECK(B.size(0) == N, "gemm_splitk: B/C N mismatch");
  int Keff = k_pad ? k_pad : K;
  TORCH_CHECK(B.size(1) == Keff, "gemm_splitk: B/K mismatch");
  TORCH_CHECK(Keff % 64 == 0 && Keff >= 128, "gemm_splitk: bad K");
  TORCH_CHECK(C2.numel() >= 3 * C.numel(),
              "gemm_splitk: C2 must hold 3 stacked partial slices");
  TORCH_CHECK(C.scalar_type() == torch::kFloat32 &&
              C2.scalar_type() == torch::kFloat32,
              "gemm_splitk: f32 partials expected");
  const float* bp = bias ? f_ptr(*bias) : nullptr;
  return launch_gemm_splitk_t<float>(bf_ptr(A), bf_ptr(B),
                                     C.data_ptr<float>(),
                                     C2.data_ptr<float>(), bp, M, N, Keff,
                                     A.size(1), B.size(1), C.size(1),
                                     current_stream());
}

static void add2_f32_bf16(const torch::Tensor& a, const torch::Tensor& b,
                          torch::Tensor& out) {
  TORCH_CHECK(a.numel() == b.numel() && a.numel() == out.numel());
  launch_add2_f32_bf16(f_ptr(a), f_ptr(b), bf_ptr_mut(out), a.numel(),
                       current_stream());
}

// out = bf16(a + sum of the first (nparts-1) stacked slices of extra)
static void addn_f32_bf16(const torch::Tensor& a, const torch::Tensor& extra,
                          torch::Tensor& out, int nparts) {
  TORCH_CHECK(nparts >= 2 && a.numel() == out.numel() &&
              extra.numel() >= (nparts - 1) * a.numel());
  launch_addn_f32_bf16(f_ptr(a), f_ptr(extra), nparts - 1, bf_ptr_mut(out),
                       a.numel(), current_stream());
}

// ---------------------------------------------------------------------------
// LSTM sequence drivers (+ hipGraph cache)
// ---------------------------------------------------------------------------
struct GraphCache {
  std::map<std::vector<uintptr_t>, hipGraphExec_t> cache;
  hipStream_t cap_stream = nullptr;

  hipStream_t capture_stream() {
    if (!cap_stream) HIP_CHECK(hipStreamCreateWithFlags(&cap_stream, hipStreamNonBlocking));
    return cap_stream;
  }
  ~GraphCache() {
    for (auto& kv : cache) (void)hipGraphExecDestroy(kv.second);
  }
};
static GraphCache g_fwd_graphs, g_bwd_graphs;
static bool g_use_graphs = true;
static bool g_use_persistent = true;
static bool g_use_persistent_bwd = false;  // see PERF.md: broadcast amplification loses

static bool g_use_fused_bwd = true;
// K-split ways for the fused backward hop (2 or 4). 4 runs ~2 blocks
// per CU, interleaving the latency-bound A/W load chains (the hop-mfma
// phase dominates the launch; census in PERF.md) — co-residency beyond
// 1 block/CU comes from occupancy, and the bounded-spin abort flag
// catches any scheduler that breaks it.
static int g_bwd_ksplit = 2;
// 2-step-batched fused backward (grid barrier + acquire between the two
// in-launch steps; see smallm_fused_bwd2_kernel). Default off pending
// the same-box A/B.
static bool g_bwd_batch2 = false;
// Threads per fused-bwd block (256 = 4 waves, 512 = 8): 8-wave blocks
// double the in-flight load streams per CU at the same 1-block/CU grid.
// Measured (profiles/s3_census_w*.txt): hop-mfma 6.32 -> 3.91 us with
// sync/dispatch flat -> bench 269K -> 289K tokens/s (+7.3%); the 4-way
// K-split alternative reached the same hop time but paid it all back in
// cross-block sync skew + 376-block dispatch. Default 8 waves.
static int g_bwd_threads = 1024;

// Fused backward step: every block of an n-tile's split group must be
// co-resident (grid ksplit*ceil(H/16): guaranteed at <= 256 blocks = 1
// per CU; relies on >= 2-blocks/CU occupancy above that) and the batch
// must fit the 32-row MFMA tile.
static bool fused_bwd_ok(int B, int H) {
  const int nbn = (H + 15) / 16;
  return g_use_fused_bwd && B <= 32 &&
         nbn * g_bwd_ksplit <= (g_bwd_ksplit == 2 ? 256 : 400);
}

// The persistent forward needs every block co-resident and one cell
// element per thread: B*HS <= 256, H even, B <= 32.
static bool persistent_ok(int B, int H) {
  return g_use_persistent && B <= 32 && (H % 2) == 0 &&
         B * persistent_hs(H) <= 256 &&
         persistent_fwd_lds(B, H) <= 160 * 1024;
}

static void lstm_seq_fwd_body(const bf16* gx, const bf16* W_h,
                              const bf16* W_pack, bf16* h_all, bf16* h_pack,
                              float* c_all, bf16* gates, bf16* rec,
                              unsigned long long* hgran,  // barrier state
                              unsigned int* abort_flag, int T, int B, int H,
                              hipStream_t stream) {
  if (persistent_ok(B, H)) {
    HIP_CHECK(hipMemsetAsync(hgran, 0, 513 * sizeof(unsigned int), stream));
    launch_lstm_persistent_fwd(gx, W_h, h_all, c_all, rec,
                               reinterpret_cast<unsigned int*>(hgran),
                               abort_flag, T, B, H, stream);
    return;
  }
  const int64_t hstep = (int64_t)B * H;
  const int64_t gstep = (int64_t)B * 4 * H;
  const int64_t pstep = (int64_t)((H + 31) / 32) * 2 * 64 * 8;
  const int HSp = persistent_hs(H);
  const int64_t rstep = (int64_t)((H + HSp - 1) / HSp) * B * 6 * HSp;
  launch_pack_a(h_all, h_pack, B, H, stream);  // slot 0 = h0
  for (int t = 0; t < T; ++t) {
    launch_lstm_cell_fwd(h_pack + t * pstep, c_all + t * hstep,
                         gx + t * gstep, W_pack, h_all + (t + 1) * hstep,
                         h_pack + (t + 1) * pstep, c_all + (t + 1) * hstep,
                         gates + t * gstep, rec + t * rstep, B, H, HSp,
                         stream);
  }
}

// h_all/c_all are [T+1, B, H] with slot 0 pre-filled with (h0, c0);
// h_pack is the zero-prefilled [T+1, KS*2*64*8] packed-h workspace;
// hgran the barrier/counter state buffer (layout in _LayerWorkspace);
// abort a u32 flag.
static void lstm_seq_fwd(const torch::Tensor& gx, const torch::Tensor& W_h,
                         const torch::Tensor& W_pack,
                         torch::Tensor& h_all, torch::Tensor& h_pack,
                         torch::Tensor& c_all, torch::Tensor& gates,
                         torch::Tensor& rec, torch::Tensor& hgran,
                         torch::Tensor& abort_flag) {
  int T = gx.size(0), B = gx.size(1);
  int H = h_all.size(2);
  TORCH_CHECK(gx.size(2) == 4 * H, "gx must be [T,B,4H]");
  // padded-barrier state spans uint32 words [0, 513)
  TORCH_CHECK(hgran.numel() * 8 >= 513 * 4,
              "lstm_seq_fwd: hgran too small for the padded barrier");
  const bf16* gxp = bf_ptr(gx);
  const bf16* whraw = bf_ptr(W_h);
  const bf16* whp = bf_ptr(W_pack);
  bf16* hp = bf_ptr_mut(h_all);
  bf16* hpk = bf_ptr_mut(h_pack);
  float* cp = f_ptr_mut(c_all);
  bf16* gp = bf_ptr_mut(gates);
  bf16* rp = bf_ptr_mut(rec);
  auto* hg = reinterpret_cast<unsigned long long*>(hgran.data_ptr());
  auto* ab = reinterpret_cast<unsigned int*>(abort_flag.data_ptr());
  auto stream = current_stream();
  // The persistent path is one memset + one kernel: no graph needed, and
  // replaying a grid-synchronized persistent kernel from a graph hangs
  // intermittently on ROCm 7.x — always run it eagerly.
  if (!g_use_graphs || persistent_ok(B, H)) {
    lstm_seq_fwd_body(gxp, whraw, whp, hp, hpk, cp, gp, rp, hg, ab, T, B, H,
                      stream);
    return;
  }
  // EVERY device pointer the captured launches bake in must key the
  // graph (advisor finding: an omitted pointer reallocated alongside
  // reused keyed ones would replay a stale graph into freed memory).
  std::vector<uintptr_t> key{(uintptr_t)gxp, (uintptr_t)whraw, (uintptr_t)whp,
                             (uintptr_t)hp, (uintptr_t)hpk, (uintptr_t)cp,
                             (uintptr_t)gp, (uintptr_t)rp, (uintptr_t)hg,
                             (uintptr_t)ab, (uintptr_t)T,
                             (uintptr_t)B, (uintptr_t)H};
  auto it = g_fwd_graphs.cache.find(key);
  if (it == g_fwd_graphs.cache.end()) {
    hipStream_t cs = g_fwd_graphs.capture_stream();
    HIP_CHECK(hipStreamBeginCapture(cs, hipStreamCaptureModeThreadLocal));
    lstm_seq_fwd_body(gxp, whraw, whp, hp, hpk, cp, gp, rp, hg, ab, T, B, H,
                      cs);
    hipGraph_t graph;
    HIP_CHECK(hipStreamEndCapture(cs, &graph));
    hipGraphExec_t exec;
    HIP_CHECK(hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0));
    HIP_CHECK(hipGraphDestroy(graph));
    it = g_fwd_graphs.cache.emplace(key, exec).first;
  }
  HIP_CHECK(hipGraphLaunch(it->second, stream));
}

static void lstm_seq_bwd_body(const bf16* dY, const bf16* gates,
                              const bf16* rec, const float* c_all,
                              const bf16* W_h_T, const bf16* WT_pack,
                              bf16* dG, bf16* dG_pack, float* dh_rec,
                              float* dc, unsigned long long* hgran,
                              unsigned int* abort_flag, int T, int B, int H,
                              hipStream_t stream) {
  if (g_use_persistent_bwd && persistent_ok(B, H) &&
      persistent_bwd_lds(B, H) <= 160 * 1024) {
    HIP_CHECK(hipMemsetAsync(hgran, 0, 513 * sizeof(unsigned int), stream));
    launch_lstm_persistent_bwd(dY, rec, W_h_T, dG, dG_pack,
                               reinterpret_cast<unsigned int*>(hgran),
                               abort_flag, T, B, H, stream);
    return;
  }
  const int64_t hstep = (int64_t)B * H;
  const int64_t gstep = (int64_t)B * 4 * H;
  const int HSp = persistent_hs(H);
  const int64_t rstep = (int64_t)((H + HSp - 1) / HSp) * B * 6 * HSp;
  float* dh2 = dh_rec + (int64_t)B * H;  // second K-slice partial
  if (fused_bwd_ok(B, H) && T >= 2) {
    // Fused path: dgate[T-1] standalone, then ONE launch per remaining
    // step doing hop[t] + dgate[t-1] (see smallm_fused_bwd_kernel).
    // dG_pack slots 0/1 alternate by step parity: a launch's phase-1
    // readers must not see its phase-2 pack writes.
    const int nbn = (H + 15) / 16;
    unsigned int* flags = reinterpret_cast<unsigned int*>(hgran) + 544;
    HIP_CHECK(hipMemsetAsync(flags, 0, nbn * sizeof(unsigned int), stream));
    const int64_t pstride = (int64_t)((4 * H + 31) / 32) * 2 * 64 * 8;
    launch_lstm_cell_bwd_elt(dY + (T - 1) * hstep, nullptr, dh2, dc,
                             rec + (T - 1) * rstep, dG + (T - 1) * gstep,
                             dG_pack, B, H, HSp, stream);
    if (g_bwd_batch2 && g_bwd_ksplit == 2 && T >= 3) {
      // 2-step-batched variant: barrier state in hgran words [0, 513)
      // (shared with the fwd barrier — sequential, re-zeroed here)
      unsigned int* pstate = reinterpret_cast<unsigned int*>(hgran);
      HIP_CHECK(hipMemsetAsync(pstate, 0, 513 * sizeof(unsigned int),
                               stream));
      int i = 0, t = T - 1;
      for (; t >= 2; t -= 2, ++i) {
        launch_smallm_fused_bwd2(
            dG_pack, WT_pack, dh_rec, dY, dc, rec, dG,
            dG_pack + pstride, dG_pack, flags, pstate, abort_flag, B, H,
            4 * H, HSp, t, (unsigned int)(2 * i + 1), (unsigned int)(i + 1),
            rstep, g_bwd_threads, stream);
      }
      if (t == 1) {  // even T: one leftover single-step launch
        launch_smallm_fused_bwd(dG_pack, WT_pack, dh_rec, dY, dc, rec, dG,
                                dG_pack + pstride, flags, abort_flag, B, H,
                                4 * H, HSp, (unsigned int)(2 * i + 1), 2,
                                g_bwd_threads, stream);
      }
      return;
    }
    for (int t = T - 1; t >= 1; --t) {
      const int i = T - 1 - t;
      bf16* rd = dG_pack + (i & 1) * pstride;
      bf16* wr = dG_pack + ((i + 1) & 1) * pstride;
      launch_smallm_fused_bwd(rd, WT_pack, dh_rec, dY + (t - 1) * hstep, dc,
                              rec + (t - 1) * rstep, dG + (t - 1) * gstep,
                              wr, flags, abort_flag, B, H, 4 * H, HSp,
                              (unsigned int)(T - t), g_bwd_ksplit,
                              g_bwd_threads, stream);
    }
    return;  // hop[0]'s dh output is unused (truncated-BPTT detach)
  }
  for (int t = T - 1; t >= 0; --t) {
    launch_lstm_cell_bwd_elt(dY + t * hstep,
                             (t == T - 1) ? nullptr : dh_rec, dh2, dc,
                             rec + t * rstep, dG + t * gstep, dG_pack,
                             B, H, HSp, stream);
    launch_smallm_packed_nt(dG_pack, WT_pack, dh_rec, dh2, B, H, 4 * H,
                            stream);
  }
}

// dc must be zero-filled by the caller before each call; dG is [T,B,4H];
// dG_pack the zero-prefilled packed-dG workspace; dh_rec a [B,H] f32
// workspace. WT_pack is the packed W_h^T shadow.
static void lstm_seq_bwd(const torch::Tensor& dY, const torch::Tensor& gates,
                         const torch::Tensor& rec, const torch::Tensor& c_all,
                         const torch::Tensor& W_h_T,
                         const torch::Tensor& WT_pack, torch::Tensor& dG,
                         torch::Tensor& dG_pack, torch::Tensor& dh_rec,
                         torch::Tensor& dc, torch::Tensor& hgran,
                         torch::Tensor& abort_flag) {
  int T = dY.size(0), B = dY.size(1), H = dY.size(2);
  const bf16* dyp = bf_ptr(dY);
  const bf16* gp = bf_ptr(gates);
  const bf16* rp = bf_ptr(rec);
  const float* cp = f_ptr(c_all);
  const bf16* whtp = bf_ptr(W_h_T);
  const bf16* wtp = bf_ptr(WT_pack);
  bf16* dgp = bf_ptr_mut(dG);
  bf16* dgpk = bf_ptr_mut(dG_pack);
  float* dhp = f_ptr_mut(dh_rec);
  float* dcp = f_ptr_mut(dc);
  auto* hg = reinterpret_cast<unsigned long long*>(hgran.data_ptr());
  auto* ab = reinterpret_cast<unsigned int*>(abort_flag.data_ptr());
  if (fused_bwd_ok(B, H) && T >= 2) {
    // pair counters live at hgran uint32[544 ..); 2 pack slots needed
    TORCH_CHECK(hgran.numel() * 8 >= 2176 + ((H + 15) / 16) * 4,
                "lstm_seq_bwd: hgran too small for fused-bwd counters");
    TORCH_CHECK(dG_pack.dim() == 2 && dG_pack.size(0) >= 2,
                "lstm_seq_bwd: fused path needs >= 2 dG_pack slots");
    TORCH_CHECK(dh_rec.numel() >= (int64_t)g_bwd_ksplit * B * H,
                "lstm_seq_bwd: dh_rec too small for ", g_bwd_ksplit,
                "-way K-split partials");
  }
  auto stream = current_stream();
  // Spin-synchronized kernels (fused pair / persistent) launch eagerly:
  // hipGraph replay of cross-block-waiting kernels hangs intermittently
  // on ROCm 7.x (PERF.md).
  if (!g_use_graphs || (fused_bwd_ok(B, H) && T >= 2) ||
      (g_use_persistent_bwd && persistent_ok(B, H) &&
       persistent_bwd_lds(B, H) <= 160 * 1024)) {
    lstm_seq_bwd_body(dyp, gp, rp, cp, whtp, wtp, dgp, dgpk, dhp, dcp, hg,
                      ab, T, B, H, stream);
    return;
  }
  // Complete pointer key (see the fwd-graph key comment).
  std::vector<uintptr_t> key{(uintptr_t)dyp, (uintptr_t)gp, (uintptr_t)rp,
                             (uintptr_t)cp, (uintptr_t)whtp,
                             (uintptr_t)wtp, (uintptr_t)dgp, (uintptr_t)dgpk,
                             (uintptr_t)dhp, (uintptr_t)dcp, (uintptr_t)hg,
                             (uintptr_t)ab, (uintptr_t)T,
                             (uintptr_t)B, (uintptr_t)H};
  auto it = g_bwd_graphs.cache.find(key);
  if (it == g_bwd_graphs.cache.end()) {
    hipStream_t cs = g_bwd_graphs.capture_stream();
    HIP_CHECK(hipStreamBeginCapture(cs, hipStreamCaptureModeThreadLocal));
    lstm_seq_bwd_body(dyp, gp, rp, cp, whtp, wtp, dgp, dgpk, dhp, dcp, hg,
                      ab, T, B, H, cs);
    hipGraph_t graph;
    HIP_CHECK(hipStreamEndCapture(cs, &graph));
    hipGraphExec_t exec;
    HIP_CHECK(hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0));
    HIP_CHECK(hipGraphDestroy(graph));
    it = g_bwd_graphs.cache.emplace(key, exec).first;
  }
  HIP_CHECK(hipGraphLaunch(it->second, stream));
}

static void set_use_graphs(bool v) { g_use_graphs = v; }
static void set_use_persistent(bool v) { g_use_persistent = v; }
static void set_use_persistent_bwd(bool v) { g_use_persistent_bwd = v; }
static void set_use_fused_bwd(bool v) { g_use_fused_bwd = v; }
static void set_bwd_ksplit(int v) {
  TORCH_CHECK(v == 2 || v == 4, "bwd ksplit must be 2 or 4");
  g_bwd_ksplit = v;
}
static void set_bwd_batch2(bool v) { g_bwd_batch2 = v; }
static void set_fwd_threads(int v) {
  TORCH_CHECK(v == 256 || v == 512, "fwd threads must be 256 or 512");
  set_fwd_threads_impl(v);
}
static void set_bwd_threads(int v) {
  TORCH_CHECK(v == 256 || v == 512 || v == 768 || v == 1024,
              "bwd threads must be 256, 512, 768 or 1024");
  g_bwd_threads = v;
}
static void clear_graphs() {
  for (auto& kv : g_fwd_graphs.cache) (void)hipGraphExecDestroy(kv.second);
  for (auto& kv : g_bwd_graphs.cache) (void)hipGraphExecDestroy(kv.second);
  g_fwd_graphs.cache.clear();
  g_bwd_graphs.cache.clear();
}

// single-step cell entry points (used by unit tests; pack on the fly)
static torch::Tensor pack_a_tmp(const torch::Tensor& A) {
  int B = A.size(0), K = A.size(1);
  int KS = (K + 31) / 32;
  auto out = torch::zeros({(int64_t)KS * 2 * 64 * 8}, A.options());
  launch_pack_a(bf_ptr(A), bf_ptr_mut(out), B, K, current_stream());
  return out;
}

static torch::Tensor pack_w_tmp(const torch::Tensor& W, int rows, int ngates,
                                int K) {
  int KS = (K + 31) / 32;
  int nb = (rows + 15) / 16;
  auto out = torch::empty({(int64_t)nb * ngates * KS * 64 * 8, }, W.options());
  launch_pack_gated_w(bf_ptr(W), bf_ptr_mut(out), rows, ngates, K,
                      current_stream());
  return out;
}

static void pack_gated_w(const torch::Tensor& W, torch::Tensor& out,
                         int64_t rows, int64_t ngates, int64_t K) {
  launch_pack_gated_w(bf_ptr(W), bf_ptr_mut(out), (int)rows, (int)ngates,
                      (int)K, current_stream());
}

static void lstm_cell_fwd_step(const torch::Tensor& h_prev,
                               const torch::Tensor& c_prev,
                               const torch::Tensor& gx,
                               const torch::Tensor& W_h, torch::Tensor& h_out,
                               torch::Tensor& c_out, torch::Tensor& gates) {
  int B = h_prev.size(0), H = h_prev.size(1);
  auto hp = pack_a_tmp(h_prev);
  auto wp = pack_w_tmp(W_h, H, 4, H);
  auto hpo = torch::zeros_like(hp);
  const int HSp = persistent_hs(H);
  const int nb = (H + HSp - 1) / HSp;
  auto rec = torch::zeros({(int64_t)nb * B * 6 * HSp}, h_prev.options());
  launch_lstm_cell_fwd(bf_ptr(hp), f_ptr(c_prev), bf_ptr(gx), bf_ptr(wp),
                       bf_ptr_mut(h_out), bf_ptr_mut(hpo), f_ptr_mut(c_out),
                       bf_ptr_mut(gates), bf_ptr_mut(rec), B, H, HSp,
                       current_stream());
}

static void smallm_gemm_nt(const torch::Tensor& A, const torch::Tensor& B,
                           torch::Tensor& C) {
  int M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K && C.size(0) == M && C.size(1) == N);
  TORCH_CHECK(M <= 32, "smallm gemm requires M <= 32");
  auto ap = pack_a_tmp(A);
  auto wp = pack_w_tmp(B, N, 1, K);
  launch_smallm_packed_nt(bf_ptr(ap), bf_ptr(wp), f_ptr_mut(C), nullptr, M,
                          N, K, current_stream());
}

// ---------------------------------------------------------------------------
// elementwise wrappers
// ---------------------------------------------------------------------------
static void embedding_fwd(const torch::Tensor& W, const torch::Tensor& idx,
                          torch::Tensor& out) {
  int N = idx.numel(), H = W.size(1);
  launch_embedding_fwd(bf_ptr(W), idx.data_ptr<int64_t>(), bf_ptr_mut(out), N,
                       H, current_stream());
}

static void embedding_bwd(const torch::Tensor& dY, const torch::Tensor& idx,
                          torch::Tensor& dW) {
  int N = idx.numel(), H = dW.size(1);
  launch_embedding_bwd(bf_ptr(dY), idx.data_ptr<int64_t>(),
                       f_ptr_mut(dW), N, H, current_stream());
}

// deterministic (fixed summation order) variant: the race-detection A/B
// oracle for the atomicAdd fast path
static void embedding_bwd_det(const torch::Tensor& dY,
                              const torch::Tensor& idx, torch::Tensor& dW) {
  int N = idx.numel(), H = dW.size(1), V = dW.size(0);
  launch_embedding_bwd_det(bf_ptr(dY), idx.data_ptr<int64_t>(),
                           f_ptr_mut(dW), N, H, V, current_stream());
}

static void dropout_fwd(const torch::Tensor& x, torch::Tensor& y, double p,
                        int64_t seed, int64_t offset) {
  launch_dropout_fwd(bf_ptr(x), bf_ptr_mut(y), (float)p, (uint64_t)seed,
                     (uint64_t)offset, x.numel(), current_stream());
}

static void dropout_bwd(const torch::Tensor& dy, torch::Tensor& dx, double p,
                        int64_t seed, int64_t offset) {
  launch_dropout_bwd(bf_ptr(dy), bf_ptr_mut(dx), (float)p, (uint64_t)seed,
                     (uint64_t)offset, dy.numel(), current_stream());
}

static void lsm_nll_fwd(const torch::Tensor& scores, const torch::Tensor& y,
                        torch::Tensor& lse, torch::Tensor& loss_accum) {
  int N = scores.size(0), V = scores.size(1);
  launch_lsm_nll_fwd(f_ptr(scores), y.data_ptr<int64_t>(), f_ptr_mut(lse),
                     f_ptr_mut(loss_accum), N, V, current_stream());
}

static void lsm_nll_bwd(const torch::Tensor& scores, const torch::Tensor& lse,
                        const torch::Tensor& y, const torch::Tensor& upstream,
                        double scale, torch::Tensor& dscores) {
  int N = scores.size(0), V = scores.size(1);
  launch_lsm_nll_bwd(f_ptr(scores), f_ptr(lse), y.data_ptr<int64_t>(),
                     f_ptr(upstream), (float)scale, f_ptr_mut(dscores), N, V,
                     current_stream());
}

static void norm2_mt(const torch::Tensor& desc, torch::Tensor& accum) {
  launch_norm2_mt(desc.data_ptr<int64_t>(), desc.size(0), f_ptr_mut(accum),
                  current_stream());
}

static void sgd_mt(const torch::Tensor& desc, const torch::Tensor& norm2,
                   double max_norm, double lr, double grad_scale) {
  launch_sgd_mt(desc.data_ptr<int64_t>(), desc.size(0), f_ptr(norm2),
                (float)max_norm, (float)lr, (float)grad_scale,
                current_stream());
}

static void norm2_accum(const torch::Tensor& g, torch::Tensor& accum) {
  launch_norm2_accum(f_ptr(g), g.numel(), f_ptr_mut(accum), current_stream());
}

static void sgd_update(torch::Tensor& master, const torch::Tensor& grad,
                       c10::optional<torch::Tensor> shadow,
                       const torch::Tensor& norm2, double max_norm, double lr,
                       double grad_scale) {
  bf16* sp = shadow ? bf_ptr_mut(*shadow) : nullptr;
  launch_sgd_update(f_ptr_mut(master), f_ptr(grad), sp, f_ptr(norm2),
                    (float)max_norm, (float)lr, (float)grad_scale,
                    master.numel(), current_stream());
}

static void colsum_bf16(const torch::Tensor& in, torch::Tensor& out) {
  int R = in.size(0), C = in.size(1);
  TORCH_CHECK(out.numel() == C);
  launch_colsum_bf16(bf_ptr(in), f_ptr_mut(out), R, C, current_stream());
}

// db = colsum(dscores) for f32 [N, V] scores grads (out pre-zeroed)
static void colsum_f32(const torch::Tensor& in, torch::Tensor& out) {
  launch_colsum_f32(f_ptr(in), f_ptr_mut(out), in.size(0), in.size(1),
                    current_stream());
}

// K13: acc += softmax(scores) rowwise (ensemble probability averaging,
// reference ensemble.py:100-105); scores/acc f32 [N, V].
static void softmax_acc(const torch::Tensor& scores, torch::Tensor& acc) {
  int N = scores.size(0), V = scores.size(1);
  TORCH_CHECK(acc.size(0) == N && acc.size(1) == V,
              "softmax_acc: acc must match scores' shape");
  launch_softmax_acc(f_ptr(scores), f_ptr_mut(acc), N, V, current_stream());
}

// dst is [C, ldd] with ldd >= R; columns [R, ldd) are left untouched
// (pre-zero them once when using the pad as a GEMM K extension).
static void transpose_bf16(const torch::Tensor& src, torch::Tensor& dst) {
  int R = src.size(0), C = src.size(1);
  int ldd = dst.size(1);
  TORCH_CHECK(dst.size(0) == C && ldd >= R);
  launch_transpose_bf16(bf_ptr(src), bf_ptr_mut(dst), R, C, ldd,
                        current_stream());
}

}  // namespace zamd

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "zaremba_amd gfx950 kernel library";
  m.def("gemm", &zamd::gemm, "MFMA bf16 GEMM (NT / TN)", py::arg("A"),
        py::arg("B"), py::arg("C"), py::arg("bias"), py::arg("trans_a"),
        py::arg("trans_b"), py::arg("k_pad") = 0);
  m.def("gemm_splitk", &zamd::gemm_splitk,
        "2-way split-K NT GEMM (f32 partials)", py::arg("A"), py::arg("B"),
        py::arg("C"), py::arg("C2"), py::arg("bias"), py::arg("k_pad") = 0);
  m.def("add2_f32_bf16", &zamd::add2_f32_bf16);
  m.def("addn_f32_bf16", &zamd::addn_f32_bf16);
  m.def("lstm_seq_fwd", &zamd::lstm_seq_fwd);
  m.def("lstm_seq_bwd", &zamd::lstm_seq_bwd);
  m.def("lstm_cell_fwd_step", &zamd::lstm_cell_fwd_step);
  m.def("pack_gated_w", &zamd::pack_gated_w);
  m.def("smallm_gemm_nt", &zamd::smallm_gemm_nt);
  m.def("embedding_fwd", &zamd::embedding_fwd);
  m.def("embedding_bwd", &zamd::embedding_bwd);
  m.def("embedding_bwd_det", &zamd::embedding_bwd_det);
  m.def("dropout_fwd", &zamd::dropout_fwd);
  m.def("dropout_bwd", &zamd::dropout_bwd);
  m.def("lsm_nll_fwd", &zamd::lsm_nll_fwd);
  m.def("lsm_nll_bwd", &zamd::lsm_nll_bwd);
  m.def("norm2_accum", &zamd::norm2_accum);
  m.def("norm2_mt", &zamd::norm2_mt);
  m.def("sgd_mt", &zamd::sgd_mt);
  m.def("sgd_update", &zamd::sgd_update);
  m.def("transpose_bf16", &zamd::transpose_bf16);
  m.def("colsum_bf16", &zamd::colsum_bf16);
  m.def("colsum_f32", &zamd::colsum_f32);
  m.def("softmax_acc", &zamd::softmax_acc);
  m.def("set_use_graphs", &zamd::set_use_graphs);
  m.def("set_use_persistent", &zamd::set_use_persistent);
  m.def("persistent_hs", &zamd::persistent_hs);
  m.def("set_use_persistent_bwd", &zamd::set_use_persistent_bwd);
  m.def("set_use_fused_bwd", &zamd::set_use_fused_bwd);
  m.def("set_bwd_ksplit", &zamd::set_bwd_ksplit);
  m.def("set_bwd_threads", &zamd::set_bwd_threads);
  m.def("set_bwd_batch2", &zamd::set_bwd_batch2);
  m.def("set_fwd_threads", &zamd::set_fwd_threads);
  m.def("fused_bwd_active", &zamd::fused_bwd_ok);
  m.def("clear_graphs", &zamd::clear_graphs);
}
