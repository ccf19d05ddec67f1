// Elementwise / reduction kernels for gfx950:
//   K1  embedding gather fwd + fp32 scatter-add bwd (reference model.py:14)
//   K5  philox inverted dropout, mask regenerated in bwd (model.py:87,105,108)
//   K7  fused log-softmax + NLL fwd/bwd (reference main.py:77-84, stable form)
//   K9/K10 fused grad-norm^2 reduce + clipped SGD with fp32 master weights
//          and bf16 shadow rewrite (reference main.py:115-117)
#include "common.h"
#include "philox.h"

namespace zamd {

// ---------------------------------------------------------------------------
// K1: embedding
// ---------------------------------------------------------------------------
__global__ void embedding_fwd_kernel(const bf16* __restrict__ W,
                                     const int64_t* __restrict__ idx,
                                     bf16* __restrict__ out, int N, int H) {
  // one wave per token row; 16B vector copies
  int row = blockIdx.x * (blockDim.x / 64) + wave_id();
  if (row >= N) return;
  const bf16* src = W + (int64_t)idx[row] * H;
  bf16* dst = out + (int64_t)row * H;
  int l = lane_id();
  int nv = H / 8;
  for (int v = l; v < nv; v += 64)
    reinterpret_cast<bf16x8*>(dst)[v] =
        reinterpret_cast<const bf16x8*>(src)[v];
  for (int k = nv * 8 + l; k < H; k += 64) dst[k] = src[k];
}

void launch_embedding_fwd(const bf16* W, const int64_t* idx, bf16* out,
                          int N, int H, hipStream_t stream) {
  int waves_per_block = 4;
  int grid = cdiv(N, waves_per_block);
  hipLaunchKernelGGL(embedding_fwd_kernel, dim3(grid),
                     dim3(waves_per_block * 64), 0, stream, W, idx, out, N, H);
}

__global__ void embedding_bwd_kernel(const bf16* __restrict__ dY,
                                     const int64_t* __restrict__ idx,
                                     float* __restrict__ dW, int N, int H) {
  int row = blockIdx.x * (blockDim.x / 64) + wave_id();
  if (row >= N) return;
  const bf16* src = dY + (int64_t)row * H;
  float* dst = dW + (int64_t)idx[row] * H;
  for (int k = lane_id(); k < H; k += 64)
    atomicAdd(dst + k, bf2f(src[k]));
}

void launch_embedding_bwd(const bf16* dY, const int64_t* idx, float* dW,
                          int N, int H, hipStream_t stream) {
  int waves_per_block = 4;
  int grid = cdiv(N, waves_per_block);
  hipLaunchKernelGGL(embedding_bwd_kernel, dim3(grid),
                     dim3(waves_per_block * 64), 0, stream, dY, idx, dW, N, H);
}

// Deterministic embedding backward (the race-detection A/B oracle,
// SURVEY.md §5): one wave per vocab row scans the index list IN ORDER
// and accumulates matching dY rows — a fixed fp32 summation order, so
// two runs are bitwise identical and the atomicAdd fast path can be
// validated against it (numerically: the atomic path reorders the same
// summands). O(V*N) index scans; validation/debug path, not the hot one.
__global__ void embedding_bwd_det_kernel(const bf16* __restrict__ dY,
                                         const int64_t* __restrict__ idx,
                                         float* __restrict__ dW, int N, int H,
                                         int V) {
  int v = blockIdx.x * (blockDim.x / 64) + wave_id();
  if (v >= V) return;
  float* dst = dW + (int64_t)v * H;
  int l = lane_id();
  for (int n = 0; n < N; ++n) {
    if ((int)idx[n] == v) {
      const bf16* src = dY + (int64_t)n * H;
      for (int k = l; k < H; k += 64) dst[k] += bf2f(src[k]);
    }
  }
}

void launch_embedding_bwd_det(const bf16* dY, const int64_t* idx, float* dW,
                              int N, int H, int V, hipStream_t stream) {
  int waves_per_block = 4;
  int grid = cdiv(V, waves_per_block);
  hipLaunchKernelGGL(embedding_bwd_det_kernel, dim3(grid),
                     dim3(waves_per_block * 64), 0, stream, dY, idx, dW, N, H,
                     V);
}

// ---------------------------------------------------------------------------
// K5: dropout (inverted, scale 1/(1-p) at train time)
// ---------------------------------------------------------------------------
// The philox offset is a HOST-side counter passed by value: dropout is
// never hipGraph-captured (capture covers only the C++ LSTM launch
// trains), so no device counter is needed — this removes a tick kernel
// + a 1-element alloc per call. The offset is saved in the autograd ctx
// and the backward regenerates the identical mask. 4 elements/philox.
__global__ void dropout_fwd_kernel(const bf16* __restrict__ x,
                                   bf16* __restrict__ y, float p,
                                   uint64_t seed, uint64_t off, int64_t n) {
  float scale = 1.f / (1.f - p);
  int64_t quad = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t nquads = (n + 3) / 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; quad < nquads; quad += stride) {
    Philox4 r = philox4x32_10(seed, off + quad);
    uint32_t u[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      int64_t i = quad * 4 + e;
      if (i < n) {
        bool keep = u32_to_uniform(u[e]) >= p;
        y[i] = keep ? f2bf(bf2f(x[i]) * scale) : (bf16)0.f;
      }
    }
  }
}

__global__ void dropout_bwd_kernel(const bf16* __restrict__ dy,
                                   bf16* __restrict__ dx, float p,
                                   uint64_t seed, uint64_t off, int64_t n) {
  float scale = 1.f / (1.f - p);
  int64_t quad = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t nquads = (n + 3) / 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; quad < nquads; quad += stride) {
    Philox4 r = philox4x32_10(seed, off + quad);
    uint32_t u[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      int64_t i = quad * 4 + e;
      if (i < n) {
        bool keep = u32_to_uniform(u[e]) >= p;
        dx[i] = keep ? f2bf(bf2f(dy[i]) * scale) : (bf16)0.f;
      }
    }
  }
}

static int dropout_grid(int64_t n) {
  int64_t nquads = (n + 3) / 4;
  int g = cdiv(nquads, 256);
  return g > 2048 ? 2048 : g;
}

void launch_dropout_fwd(const bf16* x, bf16* y, float p, uint64_t seed,
                        uint64_t offset, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(dropout_fwd_kernel, dim3(dropout_grid(n)), dim3(256), 0,
                     stream, x, y, p, seed, offset, n);
}

void launch_dropout_bwd(const bf16* dy, bf16* dx, float p, uint64_t seed,
                        uint64_t offset, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(dropout_bwd_kernel, dim3(dropout_grid(n)), dim3(256), 0,
                     stream, dy, dx, p, seed, offset, n);
}

// ---------------------------------------------------------------------------
// K7: fused log-softmax + NLL (stable; scores fp32 [N, V])
// ---------------------------------------------------------------------------
// fwd: per row, lse = max + log(sum exp(s - max)); atomically accumulates
// sum_n (lse_n - s_n[y_n]) into loss_accum (zeroed by the host wrapper);
// saves lse for backward. Loss scale (batch_size / N) applied host-side
// as a lazy torch op.
// single pass: per-thread online (max, sum) over its stripe, then a
// wave/block reduction with the online-softmax combine.
__global__ void lsm_nll_fwd_kernel(const float* __restrict__ scores,
                                   const int64_t* __restrict__ y,
                                   float* __restrict__ lse,
                                   float* __restrict__ loss_accum, int N,
                                   int V) {
  __shared__ float sm[8], ss[8];
  int row = blockIdx.x;
  if (row >= N) return;
  const float* s = scores + (int64_t)row * V;
  // Two independent online chains per thread over contiguous float4
  // halves: the original single-chain scalar loop was a serial
  // dependent update per strided load — latency-bound at ~1.2 TB/s
  // over the [700,10000] scores row; vec4 loads quarter the
  // latency-per-element and the chain pair overlaps them.
  // rows are float4-aligned only when V % 4 == 0 (s + row*V)
  const int V4 = (V & 3) == 0 ? V / 4 : 0;
  const int half4 = (V4 >= 2 * (int)blockDim.x)
                        ? (int)(((V4 + 2 * blockDim.x - 1) /
                                 (2 * blockDim.x)) * blockDim.x)
                        : V4;
  const float4* s4 = reinterpret_cast<const float4*>(s);
  float m = -INFINITY, acc = 0.f;
  float m1 = -INFINITY, a1 = 0.f;
  auto upd0 = [&](float x) {
    if (x > m) {
      acc = acc * __expf(m - x) + 1.f;
      m = x;
    } else {
      acc += __expf(x - m);
    }
  };
  auto upd1 = [&](float x) {
    if (x > m1) {
      a1 = a1 * __expf(m1 - x) + 1.f;
      m1 = x;
    } else {
      a1 += __expf(x - m1);
    }
  };
  for (int q = threadIdx.x; q < half4; q += blockDim.x) {
    float4 v = s4[q];
    const int q2 = q + half4;
    float4 u = s4[min(q2, V4 - 1)];  // clamped load; guarded update
    upd0(v.x); upd0(v.y); upd0(v.z); upd0(v.w);
    if (q2 < V4) {
      upd1(u.x); upd1(u.y); upd1(u.z); upd1(u.w);
    }
  }
  for (int v = V4 * 4 + threadIdx.x; v < V; v += blockDim.x) upd0(s[v]);
  {  // merge the two chains (guard the never-ran -inf case)
    float mn = fmaxf(m, m1);
    float pa = (m == -INFINITY) ? 0.f : acc * __expf(m - mn);
    float pb = (m1 == -INFINITY) ? 0.f : a1 * __expf(m1 - mn);
    acc = pa + pb;
    m = mn;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float m2 = __shfl_down(m, off, 64);
    float a2 = __shfl_down(acc, off, 64);
    float mn = fmaxf(m, m2);
    // guard the both-(-inf) case (threads whose stripe was empty):
    // exp(-inf - -inf) would be NaN
    float pa = (m == -INFINITY) ? 0.f : acc * __expf(m - mn);
    float pb = (m2 == -INFINITY) ? 0.f : a2 * __expf(m2 - mn);
    acc = pa + pb;
    m = mn;
  }
  const int nw = blockDim.x / 64;
  if (lane_id() == 0) {
    sm[wave_id()] = m;
    ss[wave_id()] = acc;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = sm[0], A = ss[0];
    for (int w = 1; w < nw; ++w) {
      float mn = fmaxf(M, sm[w]);
      float pa = (M == -INFINITY) ? 0.f : A * __expf(M - mn);
      float pb = (sm[w] == -INFINITY) ? 0.f : ss[w] * __expf(sm[w] - mn);
      A = pa + pb;
      M = mn;
    }
    float l = M + __logf(A);
    lse[row] = l;
    atomicAdd(loss_accum, l - s[y[row]]);
  }
}

void launch_lsm_nll_fwd(const float* scores, const int64_t* y, float* lse,
                        float* loss_accum, int N, int V, hipStream_t stream) {
  hipLaunchKernelGGL(lsm_nll_fwd_kernel, dim3(N), dim3(256), 0, stream,
                     scores, y, lse, loss_accum, N, V);
}

// bwd: dscores = (softmax - onehot_y) * gscale, gscale = upstream * B / N
// (upstream read from a device scalar so no sync is needed).
__global__ void lsm_nll_bwd_kernel(const float* __restrict__ scores,
                                   const float* __restrict__ lse,
                                   const int64_t* __restrict__ y,
                                   const float* __restrict__ upstream,
                                   float scale, float* __restrict__ dscores,
                                   int N, int V) {
  int row = blockIdx.x;
  if (row >= N) return;
  float g = *upstream * scale;
  const float* s = scores + (int64_t)row * V;
  float* d = dscores + (int64_t)row * V;
  float l = lse[row];
  int64_t yy = y[row];
  // vec4 main body; per-lane float4 keeps the wave's access one
  // contiguous 1 KB line. Rows are float4-aligned only when V % 4 == 0
  // (s + row*V) — otherwise the scalar tail loop covers everything.
  const int V4 = (V & 3) == 0 ? V / 4 : 0;
  const float4* s4 = reinterpret_cast<const float4*>(s);
  float4* d4 = reinterpret_cast<float4*>(d);
  for (int q = threadIdx.x; q < V4; q += blockDim.x) {
    float4 v = s4[q];
    const int base = q * 4;
    float4 o;
    o.x = (__expf(v.x - l) - (base + 0 == yy ? 1.f : 0.f)) * g;
    o.y = (__expf(v.y - l) - (base + 1 == yy ? 1.f : 0.f)) * g;
    o.z = (__expf(v.z - l) - (base + 2 == yy ? 1.f : 0.f)) * g;
    o.w = (__expf(v.w - l) - (base + 3 == yy ? 1.f : 0.f)) * g;
    d4[q] = o;
  }
  for (int v = V4 * 4 + threadIdx.x; v < V; v += blockDim.x) {
    float p = __expf(s[v] - l);
    d[v] = (p - (v == yy ? 1.f : 0.f)) * g;
  }
}

void launch_lsm_nll_bwd(const float* scores, const float* lse,
                        const int64_t* y, const float* upstream, float scale,
                        float* dscores, int N, int V, hipStream_t stream) {
  hipLaunchKernelGGL(lsm_nll_bwd_kernel, dim3(N), dim3(256), 0, stream,
                     scores, lse, y, upstream, scale, dscores, N, V);
}

// ---------------------------------------------------------------------------
// K13: fused softmax-accumulate for ensemble probability averaging
// (reference ensemble.py:100-105). acc[n,:] += softmax(scores[n,:]) in one
// launch per member — no per-model probability tensors, no k-way stack.
// Pass 1 per row: online (max, sumexp) as in lsm_nll_fwd; pass 2 re-reads
// the row (L2-hot, V=10k -> 40 KB) and accumulates exp(s - M) / A.
// ---------------------------------------------------------------------------
__global__ void softmax_acc_kernel(const float* __restrict__ scores,
                                   float* __restrict__ acc, int N, int V) {
  __shared__ float sm[8], ss[8];
  __shared__ float row_m, row_a;
  int row = blockIdx.x;
  if (row >= N) return;
  const float* s = scores + (int64_t)row * V;
  // vec4 online chain (see lsm_nll_fwd: the scalar strided loop was
  // load-latency bound)
  float m = -INFINITY, a = 0.f;
  auto upd = [&](float x) {
    if (x > m) {
      a = a * __expf(m - x) + 1.f;
      m = x;
    } else {
      a += __expf(x - m);
    }
  };
  // rows are float4-aligned only when V % 4 == 0 (s + row*V)
  const int V4 = (V & 3) == 0 ? V / 4 : 0;
  const float4* s4 = reinterpret_cast<const float4*>(s);
  for (int q = threadIdx.x; q < V4; q += blockDim.x) {
    float4 v = s4[q];
    upd(v.x); upd(v.y); upd(v.z); upd(v.w);
  }
  for (int v = V4 * 4 + threadIdx.x; v < V; v += blockDim.x) upd(s[v]);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float m2 = __shfl_down(m, off, 64);
    float a2 = __shfl_down(a, off, 64);
    float mn = fmaxf(m, m2);
    float pa = (m == -INFINITY) ? 0.f : a * __expf(m - mn);
    float pb = (m2 == -INFINITY) ? 0.f : a2 * __expf(m2 - mn);
    a = pa + pb;
    m = mn;
  }
  const int nw = blockDim.x / 64;
  if (lane_id() == 0) {
    sm[wave_id()] = m;
    ss[wave_id()] = a;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M = sm[0], A = ss[0];
    for (int w = 1; w < nw; ++w) {
      float mn = fmaxf(M, sm[w]);
      float pa = (M == -INFINITY) ? 0.f : A * __expf(M - mn);
      float pb = (sm[w] == -INFINITY) ? 0.f : ss[w] * __expf(sm[w] - mn);
      A = pa + pb;
      M = mn;
    }
    row_m = M;
    row_a = A;
  }
  __syncthreads();
  const float M = row_m, invA = 1.f / row_a;
  float* d = acc + (int64_t)row * V;
  float4* d4 = reinterpret_cast<float4*>(d);
  for (int q = threadIdx.x; q < V4; q += blockDim.x) {
    float4 v = s4[q];
    float4 o = d4[q];
    o.x += __expf(v.x - M) * invA;
    o.y += __expf(v.y - M) * invA;
    o.z += __expf(v.z - M) * invA;
    o.w += __expf(v.w - M) * invA;
    d4[q] = o;
  }
  for (int v = V4 * 4 + threadIdx.x; v < V; v += blockDim.x)
    d[v] += __expf(s[v] - M) * invA;
}

void launch_softmax_acc(const float* scores, float* acc, int N, int V,
                        hipStream_t stream) {
  hipLaunchKernelGGL(softmax_acc_kernel, dim3(N), dim3(256), 0, stream,
                     scores, acc, N, V);
}

// ---------------------------------------------------------------------------
// K9/K10: fused grad clip + SGD
// ---------------------------------------------------------------------------
__global__ void norm2_accum_kernel(const float* __restrict__ g, int64_t n,
                                   float* __restrict__ accum) {
  __shared__ float scratch[8];
  float acc = 0.f;
  const int64_t n4 = n / 4;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float4* g4 = reinterpret_cast<const float4*>(g);
  for (; i < n4; i += stride) {
    float4 v = g4[i];
    acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
  }
  if (blockIdx.x == 0 && threadIdx.x < (n & 3)) {
    float v = g[n4 * 4 + threadIdx.x];
    acc += v * v;
  }
  acc = block_reduce(acc, scratch, 0.f,
                     [] __device__(float a, float b) { return a + b; });
  if (threadIdx.x == 0) atomicAdd(accum, acc);
}

void launch_norm2_accum(const float* g, int64_t n, float* accum,
                        hipStream_t stream) {
  int grid = cdiv(n, 256);
  if (grid > 2048) grid = 2048;
  hipLaunchKernelGGL(norm2_accum_kernel, dim3(grid), dim3(256), 0, stream, g,
                     n, accum);
}

// ---- multi-tensor variants (one launch for all params) --------------------
// desc: per chunk of <=CHUNK_ELEMS elements, int64 fields. norm2:
// {grad_ptr, n}; sgd: {master_ptr, grad_ptr, shadow_ptr|0, n}.
constexpr int CHUNK_ELEMS = 65536;

__global__ void norm2_mt_kernel(const int64_t* __restrict__ desc,
                                float* __restrict__ accum) {
  __shared__ float scratch[8];
  const float* g = reinterpret_cast<const float*>(desc[blockIdx.x * 2]);
  const int n = (int)desc[blockIdx.x * 2 + 1];
  // 2 independent accumulator chains (the single-chain loop was a
  // serial dependent-add per strided float4 load — the same latency
  // pathology the colsum kernels had)
  float a0 = 0.f, a1 = 0.f;
  const int n4 = n / 4;
  const float4* g4 = reinterpret_cast<const float4*>(g);
  int i = threadIdx.x;
  for (; i + (int)blockDim.x < n4; i += 2 * blockDim.x) {
    float4 v = g4[i];
    float4 u = g4[i + blockDim.x];
    a0 += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    a1 += u.x * u.x + u.y * u.y + u.z * u.z + u.w * u.w;
  }
  if (i < n4) {
    float4 v = g4[i];
    a0 += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
  }
  if (threadIdx.x < (n & 3)) {
    float v = g[n4 * 4 + threadIdx.x];
    a0 += v * v;
  }
  float acc = block_reduce(a0 + a1, scratch, 0.f,
                           [] __device__(float a, float b) { return a + b; });
  if (threadIdx.x == 0) atomicAdd(accum, acc);
}

void launch_norm2_mt(const int64_t* desc, int nchunk, float* accum,
                     hipStream_t stream) {
  hipLaunchKernelGGL(norm2_mt_kernel, dim3(nchunk), dim3(256), 0, stream,
                     desc, accum);
}

__global__ void sgd_mt_kernel(const int64_t* __restrict__ desc,
                              const float* __restrict__ norm2,
                              float max_norm, float lr, float grad_scale) {
  float norm = sqrtf(*norm2) * grad_scale;
  float coef = max_norm / (norm + 1e-6f);
  coef = fminf(coef, 1.f) * lr * grad_scale;
  float* m = reinterpret_cast<float*>(desc[blockIdx.x * 4]);
  const float* g = reinterpret_cast<const float*>(desc[blockIdx.x * 4 + 1]);
  bf16* sh = reinterpret_cast<bf16*>(desc[blockIdx.x * 4 + 2]);
  const int n = (int)desc[blockIdx.x * 4 + 3];
  const int n4 = n / 4;
  float4* m4 = reinterpret_cast<float4*>(m);
  const float4* g4 = reinterpret_cast<const float4*>(g);
  bf16x4* s4 = reinterpret_cast<bf16x4*>(sh);
  for (int i = threadIdx.x; i < n4; i += blockDim.x) {
    float4 mv = m4[i];
    float4 gv = g4[i];
    mv.x -= coef * gv.x; mv.y -= coef * gv.y;
    mv.z -= coef * gv.z; mv.w -= coef * gv.w;
    m4[i] = mv;
    if (sh) s4[i] = bf16x4{f2bf(mv.x), f2bf(mv.y), f2bf(mv.z), f2bf(mv.w)};
  }
  if (threadIdx.x < (n & 3)) {
    int j = n4 * 4 + threadIdx.x;
    float v = m[j] - coef * g[j];
    m[j] = v;
    if (sh) sh[j] = f2bf(v);
  }
}

void launch_sgd_mt(const int64_t* desc, int nchunk, const float* norm2,
                   float max_norm, float lr, float grad_scale,
                   hipStream_t stream) {
  hipLaunchKernelGGL(sgd_mt_kernel, dim3(nchunk), dim3(256), 0, stream, desc,
                     norm2, max_norm, lr, grad_scale);
}

// master -= lr * grad_scale * clip(norm) * grad; shadow/bf16 rewritten in
// the same pass. `shadow` may be null (biases stay fp32).
__global__ void sgd_update_kernel(float* __restrict__ master,
                                  const float* __restrict__ grad,
                                  bf16* __restrict__ shadow,
                                  const float* __restrict__ norm2,
                                  float max_norm, float lr, float grad_scale,
                                  int64_t n) {
  float norm = sqrtf(*norm2) * grad_scale;
  float coef = max_norm / (norm + 1e-6f);
  coef = fminf(coef, 1.f) * lr * grad_scale;
  const int64_t n4 = n / 4;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float4* m4 = reinterpret_cast<float4*>(master);
  const float4* g4 = reinterpret_cast<const float4*>(grad);
  bf16x4* s4 = reinterpret_cast<bf16x4*>(shadow);
  for (; i < n4; i += stride) {
    float4 m = m4[i];
    float4 g = g4[i];
    m.x -= coef * g.x; m.y -= coef * g.y;
    m.z -= coef * g.z; m.w -= coef * g.w;
    m4[i] = m;
    if (shadow) s4[i] = bf16x4{f2bf(m.x), f2bf(m.y), f2bf(m.z), f2bf(m.w)};
  }
  if (blockIdx.x == 0 && threadIdx.x < (n & 3)) {
    int64_t j = n4 * 4 + threadIdx.x;
    float v = master[j] - coef * grad[j];
    master[j] = v;
    if (shadow) shadow[j] = f2bf(v);
  }
}

void launch_sgd_update(float* master, const float* grad, bf16* shadow,
                       const float* norm2, float max_norm, float lr,
                       float grad_scale, int64_t n, hipStream_t stream) {
  int grid = cdiv(n, 256);
  if (grid > 2048) grid = 2048;
  hipLaunchKernelGGL(sgd_update_kernel, dim3(grid), dim3(256), 0, stream,
                     master, grad, shadow, norm2, max_norm, lr, grad_scale, n);
}

// transposed bf16 shadow refresh: dst[c][r] = src[r][c].
// 64x64 LDS tile, 16-B vector loads AND stores; interior blocks take a
// guard-free path (per-element guards around global loads serialize).
__global__ void transpose_bf16_kernel(const bf16* __restrict__ src,
                                      bf16* __restrict__ dst, int R, int C,
                                      int ldd) {  // dst row stride (>= R)
  __shared__ bf16 tile[64][72];  // +8 bf16 row pad (16 B): conflict relief
  const int c0 = blockIdx.x * 64, r0 = blockIdx.y * 64;
  const int t = threadIdx.x;
  const int lr = t / 8;           // 0..31 (row within pass)
  const int lc8 = (t % 8) * 8;    // 0..56
  const bool interior = (r0 + 64 <= R) && (c0 + 64 <= C);
  if (interior) {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int r = lr + p * 32;
      *reinterpret_cast<bf16x8*>(&tile[r][lc8]) =
          *reinterpret_cast<const bf16x8*>(src + (int64_t)(r0 + r) * C + c0 +
                                           lc8);
    }
  } else {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int r = lr + p * 32;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int rr = r0 + r, cc = c0 + lc8 + e;
        tile[r][lc8 + e] =
            (rr < R && cc < C) ? src[(int64_t)rr * C + cc] : (bf16)0.f;
      }
    }
  }
  __syncthreads();
  if (interior) {
    // write: thread covers TWO ADJACENT dst rows (c0+cp, c0+cp+1), cols
    // r0+lc8..+8 — the adjacent pair turns the scattered per-element
    // LDS reads into 4-B paired reads (half the ds_read instructions of
    // the old one-col-per-pass scheme)
    const int cp = (threadIdx.x / 8) * 2;
    bf16x8 v0, v1;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const unsigned int pr = *reinterpret_cast<const unsigned int*>(
          &tile[lc8 + e][cp]);
      v0[e] = __builtin_bit_cast(bf16, (unsigned short)(pr & 0xffffu));
      v1[e] = __builtin_bit_cast(bf16, (unsigned short)(pr >> 16));
    }
    *reinterpret_cast<bf16x8*>(dst + (int64_t)(c0 + cp) * ldd + r0 + lc8) =
        v0;
    *reinterpret_cast<bf16x8*>(dst + (int64_t)(c0 + cp + 1) * ldd + r0 +
                               lc8) = v1;
  } else {
#pragma unroll
    for (int p = 0; p < 2; ++p) {
      const int c = lr + p * 32;   // source col == dst row offset
      bf16x8 v;
#pragma unroll
      for (int e = 0; e < 8; ++e) v[e] = tile[lc8 + e][c];
      if (c0 + c < C) {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          if (r0 + lc8 + e < R)
            dst[(int64_t)(c0 + c) * ldd + r0 + lc8 + e] = v[e];
      }
    }
  }
}

void launch_transpose_bf16(const bf16* src, bf16* dst, int R, int C,
                           int ldd, hipStream_t stream) {
  dim3 grid(cdiv(C, 64), cdiv(R, 64));
  hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0, stream, src,
                     dst, R, C, ldd);
}

// column sum of a bf16 matrix -> f32 (bias grads: db = colsum(dG)).
// 2-D grid: row-chunks in y accumulate via atomicAdd (out pre-zeroed by
// the wrapper); a single serial column walk was latency-bound (165 us).
// 4 independent accumulators: the single-acc loop was a serial
// dependent-add chain over C-strided (cache-line-apart) loads —
// 1.65 TB/s; four chains overlap the load latencies.
__global__ void colsum_bf16_kernel(const bf16* __restrict__ in,
                                   float* __restrict__ out, int Rr, int Cc) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= Cc) return;
  int r0 = blockIdx.y * 64;
  int r1 = min(r0 + 64, Rr);
  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
  int r = r0;
  for (; r + 4 <= r1; r += 4) {
    a0 += bf2f(in[(int64_t)(r + 0) * Cc + c]);
    a1 += bf2f(in[(int64_t)(r + 1) * Cc + c]);
    a2 += bf2f(in[(int64_t)(r + 2) * Cc + c]);
    a3 += bf2f(in[(int64_t)(r + 3) * Cc + c]);
  }
  for (; r < r1; ++r) a0 += bf2f(in[(int64_t)r * Cc + c]);
  atomicAdd(out + c, (a0 + a1) + (a2 + a3));
}

void launch_colsum_bf16(const bf16* in, float* out, int R, int C,
                        hipStream_t stream) {
  dim3 grid(cdiv(C, 256), cdiv(R, 64));
  hipLaunchKernelGGL(colsum_bf16_kernel, grid, dim3(256), 0, stream, in, out,
                     R, C);
}

// f32 column sum (projection db = colsum(dscores), f32 [N,V] input):
// same row-chunked atomicAdd scheme + 4-chain ILP as the bf16 variant.
__global__ void colsum_f32_kernel(const float* __restrict__ in,
                                  float* __restrict__ out, int Rr, int Cc) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= Cc) return;
  int r0 = blockIdx.y * 64;
  int r1 = min(r0 + 64, Rr);
  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
  int r = r0;
  for (; r + 4 <= r1; r += 4) {
    a0 += in[(int64_t)(r + 0) * Cc + c];
    a1 += in[(int64_t)(r + 1) * Cc + c];
    a2 += in[(int64_t)(r + 2) * Cc + c];
    a3 += in[(int64_t)(r + 3) * Cc + c];
  }
  for (; r < r1; ++r) a0 += in[(int64_t)r * Cc + c];
  atomicAdd(out + c, (a0 + a1) + (a2 + a3));
}

void launch_colsum_f32(const float* in, float* out, int R, int C,
                       hipStream_t stream) {
  dim3 grid(cdiv(C, 256), cdiv(R, 64));
  hipLaunchKernelGGL(colsum_f32_kernel, grid, dim3(256), 0, stream, in, out,
                     R, C);
}

// combine the two f32 partials of a split-K GEMM into the bf16 result
// (one rounding: partials carry the exact f32 MFMA accumulators).
__global__ void add2_f32_bf16_kernel(const float* __restrict__ a,
                                     const float* __restrict__ b,
                                     bf16* __restrict__ out, int64_t n) {
  int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (i + 8 <= n) {
    float4 a0 = *(const float4*)(a + i), a1 = *(const float4*)(a + i + 4);
    float4 b0 = *(const float4*)(b + i), b1 = *(const float4*)(b + i + 4);
    bf16x8 v = {(bf16)(a0.x + b0.x), (bf16)(a0.y + b0.y),
                (bf16)(a0.z + b0.z), (bf16)(a0.w + b0.w),
                (bf16)(a1.x + b1.x), (bf16)(a1.y + b1.y),
                (bf16)(a1.z + b1.z), (bf16)(a1.w + b1.w)};
    *(bf16x8*)(out + i) = v;
  } else {
    for (; i < n; ++i) out[i] = (bf16)(a[i] + b[i]);
  }
}

void launch_add2_f32_bf16(const float* a, const float* b, bf16* out,
                          int64_t n, hipStream_t stream) {
  int64_t thr = cdiv(n, 8);
  hipLaunchKernelGGL(add2_f32_bf16_kernel, dim3(cdiv(thr, 256)), dim3(256),
                     0, stream, a, b, out, n);
}

// combine a + the nextra stacked [n] f32 partial slices of `extra` into
// bf16 (N-way split-K; exact MFMA accumulators, one rounding)
__global__ void addn_f32_bf16_kernel(const float* __restrict__ a,
                                     const float* __restrict__ extra,
                                     int nextra, bf16* __restrict__ out,
                                     int64_t n) {
  int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i + 4 <= n) {
    float4 v = *(const float4*)(a + i);
    for (int p = 0; p < nextra; ++p) {
      float4 e = *(const float4*)(extra + (int64_t)p * n + i);
      v.x += e.x; v.y += e.y; v.z += e.z; v.w += e.w;
    }
    *(bf16x4*)(out + i) =
        bf16x4{f2bf(v.x), f2bf(v.y), f2bf(v.z), f2bf(v.w)};
  } else {
    for (; i < n; ++i) {
      float v = a[i];
      for (int p = 0; p < nextra; ++p) v += extra[(int64_t)p * n + i];
      out[i] = f2bf(v);
    }
  }
}

void launch_addn_f32_bf16(const float* a, const float* extra, int nextra,
                          bf16* out, int64_t n, hipStream_t stream) {
  int64_t thr = cdiv(n, 4);
  hipLaunchKernelGGL(addn_f32_bf16_kernel, dim3(cdiv(thr, 256)), dim3(256),
                     0, stream, a, extra, nextra, out, n);
}

}  // namespace zamd
