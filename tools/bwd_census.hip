// Phase census for the fused backward step kernel: compiles the REAL
// kernel (zaremba_amd/csrc/lstm.hip) with -DZAMD_BWD_PROF and drives the
// same T-step launch train as the sequence driver. Reports mean/min/max
// per phase per launch — attribution for the 12 us/launch fused-bwd
// cost (PERF.md round-2 lever).
//
// Build: hipcc --offload-arch=gfx950 -O3 -DZAMD_BWD_PROF \
//          tools/bwd_census.hip -o tools/bwd_census
// Run (GPU box): ./tools/bwd_census [H=1500] [T=35] [B=20] [iters=20]
#include "../zaremba_amd/csrc/lstm.hip"

#include <cstdio>
#include <cstdlib>
#include <vector>

#define CHK(x)                                                      \
  do {                                                              \
    hipError_t e_ = (x);                                            \
    if (e_ != hipSuccess) {                                         \
      fprintf(stderr, "HIP error %s at line %d\n",                  \
              hipGetErrorString(e_), __LINE__);                     \
      exit(1);                                                      \
    }                                                               \
  } while (0)

int main(int argc, char** argv) {
  int H = argc > 1 ? atoi(argv[1]) : 1500;
  int T = argc > 2 ? atoi(argv[2]) : 35;
  int B = argc > 3 ? atoi(argv[3]) : 20;
  int iters = argc > 4 ? atoi(argv[4]) : 20;
  int nsplit = argc > 5 ? atoi(argv[5]) : 2;  // K-split ways (2 or 4)
  int nthr = argc > 6 ? atoi(argv[6]) * 64 : 256;  // waves/block (4 or 8)
  using zamd::bf16;
  const int HSp = 8;  // persistent_hs(H) for H >= 1500-ish; rec layout only
  const int K = 4 * H;
  const int KS = (K + 31) / 32;
  const int nbn = (H + 15) / 16;
  const size_t pslot = (size_t)KS * 2 * 64 * 8;
  const size_t rslot = (size_t)((H + HSp - 1) / HSp) * B * 6 * HSp;

  bf16 *A0, *A1, *W_pack, *dY, *rec, *dG;
  float *P, *dc;
  unsigned int *flags, *abortf;
  CHK(hipMalloc(&A0, pslot * 2));
  CHK(hipMalloc(&A1, pslot * 2));
  CHK(hipMalloc(&W_pack, (size_t)nbn * KS * 64 * 8 * 2));
  CHK(hipMalloc(&dY, (size_t)T * B * H * 2));
  CHK(hipMalloc(&rec, rslot * T * 2));
  CHK(hipMalloc(&dG, (size_t)T * B * K * 2));
  CHK(hipMalloc(&P, (size_t)4 * B * H * 4));
  CHK(hipMalloc(&dc, (size_t)B * H * 4));
  CHK(hipMalloc(&flags, (size_t)nbn * 4));
  CHK(hipMalloc(&abortf, 4));
  // small nonzero bf16 payloads
  {
    std::vector<unsigned short> w((size_t)nbn * KS * 64 * 8, 0x3c00);
    CHK(hipMemcpy(W_pack, w.data(), w.size() * 2, hipMemcpyHostToDevice));
  }
  CHK(hipMemset(A0, 0x3c, pslot * 2));
  CHK(hipMemset(A1, 0x3c, pslot * 2));
  CHK(hipMemset(dY, 0x3b, (size_t)T * B * H * 2));
  CHK(hipMemset(rec, 0x3b, rslot * T * 2));
  CHK(hipMemset(dc, 0, (size_t)B * H * 4));
  CHK(hipMemset(abortf, 0, 4));

  auto run_train = [&]() {
    CHK(hipMemset(flags, 0, (size_t)nbn * 4));
    for (int t = T - 1; t >= 1; --t) {
      const int i = T - 1 - t;
      bf16* rd = (i & 1) ? A1 : A0;
      bf16* wr = (i & 1) ? A0 : A1;
      zamd::launch_smallm_fused_bwd(
          rd, W_pack, P, dY + (size_t)(t - 1) * B * H, dc,
          rec + (size_t)(t - 1) * rslot, dG + (size_t)(t - 1) * B * K, wr,
          flags, abortf, B, H, K, HSp, (unsigned int)(T - t), nsplit, nthr,
          nullptr);
    }
  };

  for (int i = 0; i < 3; ++i) run_train();
  CHK(hipDeviceSynchronize());
  static unsigned long long zero[512 * 8];
  CHK(hipMemcpyToSymbol(HIP_SYMBOL(zamd::g_bwd_prof), zero, sizeof(zero)));
  // reset the launch-gap chain so the first timed launch (whose "gap"
  // would span the warmup sync) is skipped
  CHK(hipMemcpyToSymbol(HIP_SYMBOL(zamd::g_bwd_prev_exit), zero,
                        sizeof(unsigned long long)));

  hipEvent_t e0, e1;
  CHK(hipEventCreate(&e0));
  CHK(hipEventCreate(&e1));
  CHK(hipEventRecord(e0, nullptr));
  for (int i = 0; i < iters; ++i) run_train();
  CHK(hipEventRecord(e1, nullptr));
  CHK(hipDeviceSynchronize());
  float wall_ms = 0.f;
  CHK(hipEventElapsedTime(&wall_ms, e0, e1));
  unsigned int ab = 0;
  CHK(hipMemcpy(&ab, abortf, 4, hipMemcpyDeviceToHost));

  static unsigned long long prof[512 * 8];
  CHK(hipMemcpyFromSymbol(prof, HIP_SYMBOL(zamd::g_bwd_prof), sizeof(prof)));

  const char* names[6] = {"hop-mfma",  "prefetch+red", "publish+drain",
                          "pair-sync", "dgate+stores", "entry-preamble"};
  const double launches = (double)iters * (T - 1);
  const int grid = nbn * nsplit;
  printf("H=%d T=%d B=%d grid=%d ksplit=%d thr=%d iters=%d abort=%u\n", H, T,
         B, grid, nsplit, nthr, iters, ab);
  printf("wall: %.3f us/launch (%d launches/train)\n",
         wall_ms * 1000.0 / iters / (T - 1), T - 1);
  printf("%-14s %9s %9s %9s   (us/launch)\n", "phase", "mean", "min", "max");
  double tot = 0;
  for (int p = 0; p < 6; ++p) {
    double mn = 1e30, mx = 0, sum = 0;
    for (int b = 0; b < grid; ++b) {
      double v = (double)prof[b * 8 + p] * 0.01 / launches;
      mn = v < mn ? v : mn;
      mx = v > mx ? v : mx;
      sum += v;
    }
    if (p < 5) tot += sum / grid;  // preamble overlaps phase accounting
    printf("%-14s %9.3f %9.3f %9.3f\n", names[p], sum / grid, mn, mx);
  }
  printf("phase total    %9.3f  (excl. entry-preamble)\n", tot);
  // block 0's exit->entry chain across consecutive launches: the pure
  // dispatch / launch-boundary cost (first launch of each train spans
  // the flags memset; the first timed launch is skipped)
  printf("launch-gap(b0) %9.3f  (exit[k-1] -> entry[k])\n",
         (double)prof[6] * 0.01 / (launches - 1));
  return 0;
}
