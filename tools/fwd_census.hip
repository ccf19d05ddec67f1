// Phase census for the persistent LSTM forward: compiles the REAL
// kernel (zaremba_amd/csrc/lstm_persistent.hip) with -DZAMD_FWD_PROF so
// every phase accumulates s_memrealtime (100 MHz) deltas per block, then
// reports mean/min/max per phase per timestep-layer. Attribution for the
// 14+ us/step-layer forward cost (PERF.md round-2 lever).
//
// Build: hipcc --offload-arch=gfx950 -O3 -DZAMD_FWD_PROF \
//          tools/fwd_census.hip -o tools/fwd_census
// Run (GPU box): ./tools/fwd_census [H=1500] [T=35] [B=20] [iters=50]
#include "../zaremba_amd/csrc/lstm_persistent.hip"

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
  int iters = argc > 4 ? atoi(argv[4]) : 50;
  using zamd::bf16;
  const int HS = zamd::persistent_hs(H);
  const int NB = (H + HS - 1) / HS;

  bf16 *gx, *h_all, *rec;
  float* c_all;
  unsigned int *pstate, *abort_flag;
  CHK(hipMalloc(&gx, (size_t)T * B * 4 * H * 2));
  CHK(hipMalloc(&h_all, (size_t)(T + 1) * B * H * 2));
  CHK(hipMalloc(&c_all, (size_t)(T + 1) * B * H * 4));
  CHK(hipMalloc(&rec, (size_t)T * NB * B * 6 * HS * 2));
  CHK(hipMalloc(&pstate, 64));
  CHK(hipMalloc(&abort_flag, 4));
  bf16* W_h;
  CHK(hipMalloc(&W_h, (size_t)4 * H * H * 2));
  // small nonzero values (avoid NaN/denormal timing artifacts)
  {
    std::vector<unsigned short> host((size_t)4 * H * H, 0x3c00 /* ~0.0078 */);
    CHK(hipMemcpy(W_h, host.data(), host.size() * 2, hipMemcpyHostToDevice));
    std::vector<unsigned short> hg((size_t)T * B * 4 * H, 0x3b80);
    CHK(hipMemcpy(gx, hg.data(), hg.size() * 2, hipMemcpyHostToDevice));
  }
  CHK(hipMemset(h_all, 0, (size_t)(T + 1) * B * H * 2));
  CHK(hipMemset(c_all, 0, (size_t)(T + 1) * B * H * 4));
  CHK(hipMemset(abort_flag, 0, 4));
  CHK(hipMemsetD32(hipDeviceptr_t(pstate), 0, 16));

  // warmup
  for (int i = 0; i < 5; ++i) {
    CHK(hipMemsetD32(hipDeviceptr_t(pstate), 0, 16));
    zamd::launch_lstm_persistent_fwd(gx, W_h, h_all, c_all, rec, pstate,
                                     abort_flag, T, B, H, nullptr);
  }
  CHK(hipDeviceSynchronize());
  // zero the profile accumulator
  unsigned long long zero[256 * 8] = {};
  CHK(hipMemcpyToSymbol(HIP_SYMBOL(zamd::g_fwd_prof), zero, sizeof(zero)));

  hipEvent_t e0, e1;
  CHK(hipEventCreate(&e0));
  CHK(hipEventCreate(&e1));
  CHK(hipEventRecord(e0, nullptr));
  for (int i = 0; i < iters; ++i) {
    CHK(hipMemsetD32(hipDeviceptr_t(pstate), 0, 16));
    zamd::launch_lstm_persistent_fwd(gx, W_h, h_all, c_all, rec, pstate,
                                     abort_flag, T, B, H, nullptr);
  }
  CHK(hipEventRecord(e1, nullptr));
  CHK(hipDeviceSynchronize());
  float wall_ms = 0.f;
  CHK(hipEventElapsedTime(&wall_ms, e0, e1));
  unsigned int ab = 0;
  CHK(hipMemcpy(&ab, abort_flag, 4, hipMemcpyDeviceToHost));

  static unsigned long long prof[256 * 8];
  CHK(hipMemcpyFromSymbol(prof, HIP_SYMBOL(zamd::g_fwd_prof), sizeof(prof)));

  const char* names[6] = {"gx-prefetch", "barrier",  "stage-h",
                          "mfma+gbuf",   "pointwise", "publish"};
  const double steps = (double)iters * T;
  printf("H=%d T=%d B=%d NB=%d iters=%d abort=%u\n", H, T, B, NB, iters, ab);
  printf("wall: %.2f us/launch = %.3f us/step-layer\n",
         wall_ms * 1000.0 / iters, wall_ms * 1000.0 / iters / T);
  printf("%-12s %9s %9s %9s   (us/step-layer, 100 MHz ticks x10ns)\n",
         "phase", "mean", "min", "max");
  double tot = 0;
  for (int p = 0; p < 6; ++p) {
    double mn = 1e30, mx = 0, sum = 0;
    for (int b = 0; b < NB; ++b) {
      double v = (double)prof[b * 8 + p] * 0.01 / steps;  // us
      mn = v < mn ? v : mn;
      mx = v > mx ? v : mx;
      sum += v;
    }
    tot += sum / NB;
    printf("%-12s %9.3f %9.3f %9.3f\n", names[p], sum / NB, mn, mx);
  }
  printf("phase total  %9.3f (vs wall %.3f; gap = loop/launch overhead)\n",
         tot, wall_ms * 1000.0 / iters / T);
  return 0;
}
