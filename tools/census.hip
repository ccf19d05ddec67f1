// Residency census + grid-barrier probe for the persistent LSTM kernel
// geometry (256 threads, ~141 KB dynamic LDS). Prints, for a range of
// grid sizes, whether all blocks were co-resident (each arrives at a
// counter and spins until it reads the full count, with a bounded spin).
//
// Build: hipcc --offload-arch=gfx950 -O2 tools/census.hip -o tools/census
#include <hip/hip_runtime.h>

#include <cstdio>

#define RLX_AGENT __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT
typedef __attribute__((address_space(1))) unsigned int gu32;

__global__ void census_kernel(unsigned int* cnt, unsigned int* result,
                              int nb) {
  extern __shared__ char smem[];
  smem[threadIdx.x] = 0;  // touch LDS so it is really allocated
  if (threadIdx.x == 0) {
    gu32* c = (gu32*)(uintptr_t)cnt;
    __hip_atomic_fetch_add(c, 1u, RLX_AGENT);
    unsigned int spins = 0;
    while (__hip_atomic_load(c, RLX_AGENT) < (unsigned int)nb) {
      __builtin_amdgcn_s_sleep(8);
      if (++spins > 3000000u) {
        atomicAdd(result + 1, 1u);  // timed out
        return;
      }
    }
    atomicAdd(result, 1u);  // saw the full count
  }
}

// the same two-level barrier as lstm_persistent.hip, iterated
template <int SLEEP, int VARIANT>  // VARIANT 0: two-level; 1: flat top-counter poll; 2: two-level no-acquire (timing only); 3: two-level PADDED (each group counter/gen on its own 128B line) no-acquire; 4: flat no-acquire
__device__ bool xcd_barrier(unsigned int* pstate, int grp, int nbg,
                            int ngroups, int nb, unsigned int gen,
                            unsigned int* fail) {
  __shared__ int ok_s;
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    ok_s = 1;
    gu32* st = (gu32*)(uintptr_t)pstate;
    if (VARIANT == 1 || VARIANT == 4) {
      __hip_atomic_fetch_add(&st[8], 1u, RLX_AGENT);
      unsigned int spins = 0;
      while (__hip_atomic_load(&st[8], RLX_AGENT) < gen * nb) {
        __builtin_amdgcn_s_sleep(SLEEP);
        if (++spins > 3000000u) { atomicAdd(fail, 1u); ok_s = 0; break; }
      }
      if (VARIANT == 1)
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    } else if (VARIANT == 3) {
      // padded two-level: group counter at st[grp*32], top at st[256],
      // per-group generation at st[288 + grp*32] — every hot word on its
      // own 128-B line, so the 8 group add-chains run in parallel
      unsigned int t = __hip_atomic_fetch_add(&st[512 + grp * 32], 1u, RLX_AGENT);
      if (t == gen * nbg - 1) {
        unsigned int tt = __hip_atomic_fetch_add(&st[768 + 256], 1u, RLX_AGENT);
        if (tt == gen * ngroups - 1) {
          for (int x = 0; x < 8; ++x)
            __hip_atomic_store(&st[1056 + x * 32], gen, RLX_AGENT);
        }
      }
      unsigned int spins = 0;
      while (__hip_atomic_load(&st[1056 + grp * 32], RLX_AGENT) < gen) {
        __builtin_amdgcn_s_sleep(SLEEP);
        if (++spins > 3000000u) { atomicAdd(fail, 1u); ok_s = 0; break; }
      }
    } else {
      unsigned int t = __hip_atomic_fetch_add(&st[grp], 1u, RLX_AGENT);
      if (t == gen * nbg - 1) {
        unsigned int tt = __hip_atomic_fetch_add(&st[8], 1u, RLX_AGENT);
        if (tt == gen * ngroups - 1) {
          for (int x = 0; x < 8; ++x)
            __hip_atomic_store(&st[9 + x], gen, RLX_AGENT);
        }
      }
      unsigned int spins = 0;
      while (__hip_atomic_load(&st[9 + grp], RLX_AGENT) < gen) {
        __builtin_amdgcn_s_sleep(SLEEP);
        if (++spins > 3000000u) { atomicAdd(fail, 1u); ok_s = 0; break; }
      }
      if (VARIANT != 2) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    }
  }
  __syncthreads();
  return ok_s != 0;
}

template <int SLEEP, int VARIANT>
__global__ void barrier_probe_kernel(unsigned int* pstate,
                                     unsigned int* fail, unsigned int* done,
                                     int iters, int nb) {
  extern __shared__ char smem[];
  smem[threadIdx.x] = 0;
  const int grp = blockIdx.x & 7;
  const int ngroups = nb < 8 ? nb : 8;
  const int nbg = (nb - grp + 7) / 8;
  for (int t = 1; t <= iters; ++t) {
    if (!xcd_barrier<SLEEP, VARIANT>(pstate, grp, nbg, ngroups, nb,
                                     (unsigned int)t, fail))
      return;
  }
  if (threadIdx.x == 0) atomicAdd(done, 1u);
}

int main() {
  size_t lds = (size_t)(4 * 6 + 20) * 1512 * 2 + 4 * 32 * 16 * 4;
  printf("LDS request: %zu bytes\n", lds);
  unsigned int* buf;
  (void)hipMalloc(&buf, 16384);
  for (int nb : {128, 200, 240, 248, 250, 252, 256, 260}) {
    (void)hipMemset(buf, 0, 16384);
    hipLaunchKernelGGL(census_kernel, dim3(nb), dim3(256), lds, 0, buf,
                       buf + 64, nb);
    (void)hipDeviceSynchronize();
    unsigned int res[2];
    (void)hipMemcpy(res, buf + 64, 8, hipMemcpyDeviceToHost);
    printf("census nb=%3d: full=%u timeout=%u\n", nb, res[0], res[1]);
  }
  // barrier probe at the persistent-kernel sizes (timed)
  for (int nb : {188, 125, 94}) {
    auto run = [&](const char* name, auto kern) {
      (void)hipMemset(buf, 0, 16384);
      hipEvent_t e0, e1;
      (void)hipEventCreate(&e0);
      (void)hipEventCreate(&e1);
      (void)hipEventRecord(e0, 0);
      hipLaunchKernelGGL(kern, dim3(nb), dim3(256), lds, 0, buf, buf + 64,
                         buf + 65, 1000, nb);
      (void)hipEventRecord(e1, 0);
      (void)hipDeviceSynchronize();
      float ms = 0;
      (void)hipEventElapsedTime(&ms, e0, e1);
      unsigned int res[2];
      (void)hipMemcpy(res, buf + 64, 8, hipMemcpyDeviceToHost);
      printf("barrier nb=%3d %-22s: fail=%u done=%u  %.2f us/barrier\n", nb,
             name, res[0], res[1], ms * 1000.f / 1000);
    };
    run("2lvl sleep8", barrier_probe_kernel<8, 0>);
    run("2lvl sleep2", barrier_probe_kernel<2, 0>);
    run("2lvl sleep32", barrier_probe_kernel<32, 0>);
    run("flat sleep8", barrier_probe_kernel<8, 1>);
    run("2lvl sleep8 noacq", barrier_probe_kernel<8, 2>);
    run("flat sleep8 noacq", barrier_probe_kernel<8, 4>);
    run("flat sleep2 noacq", barrier_probe_kernel<2, 4>);
    run("PADDED2lvl sl8 noacq", barrier_probe_kernel<8, 3>);
    run("PADDED2lvl sl2 noacq", barrier_probe_kernel<2, 3>);
    run("PADDED2lvl sl1 noacq", barrier_probe_kernel<1, 3>);
  }
  return 0;
}
