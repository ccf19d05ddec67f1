// Common device-side helpers for the zaremba_amd CDNA4 (gfx950) kernels.
//
// Everything here is written MI355X-first: wave64, MFMA bf16 fragments
// with fp32 accumulation, 16-byte vector memory access. No CUDA-compat
// paths. See /root/repo/SURVEY.md §2.3 for the kernel plan this
// implements (K1-K14 mapped from the reference's implicit ATen call
// sites; reference files cited per kernel).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

#define DEV_INLINE __device__ __forceinline__

namespace zamd {

using bf16 = __bf16;
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short s16x8 __attribute__((ext_vector_type(8)));

constexpr int WAVE = 64;

DEV_INLINE int lane_id() { return threadIdx.x & (WAVE - 1); }
DEV_INLINE int wave_id() { return threadIdx.x / WAVE; }

constexpr int cdiv_const(int a, int b) { return (a + b - 1) / b; }
inline int cdiv(int64_t a, int64_t b) { return (int)((a + b - 1) / b); }

// ---- MFMA wrappers (gfx950: 2xK bf16 shapes) ------------------------------
DEV_INLINE f32x4 mfma_16x16x32_bf16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// ---- conversions -----------------------------------------------------------
DEV_INLINE float bf2f(bf16 v) { return (float)v; }
DEV_INLINE bf16 f2bf(float v) { return (bf16)v; }

// ---- wave/block reductions -------------------------------------------------
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;  // valid in lane 0
}

DEV_INLINE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}

// Block reduction via LDS; `scratch` must hold >= blockDim.x/64 floats.
template <typename Op>
DEV_INLINE float block_reduce(float v, float* scratch, float init, Op op) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = op(v, __shfl_down(v, off, 64));
  const int nw = blockDim.x / 64;
  if (lane_id() == 0) scratch[wave_id()] = v;
  __syncthreads();
  float r = init;
  if (threadIdx.x < nw) r = scratch[threadIdx.x];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) r = op(r, __shfl_down(r, off, 64));
  r = __shfl(r, 0, 64);
  return r;  // valid in every lane of wave 0; callers broadcast via LDS if needed
}

}  // namespace zamd
