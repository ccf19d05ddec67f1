// Host-side launcher prototypes for the gfx950 kernel library.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace zamd {

using bf16 = __bf16;

// gemm.hip — C[M,N] = A' @ B' (+bias); TRANS_A: A is [K,M]; TRANS_B: B is
// [K,N] (otherwise the NT weight layout B[N,K]).
template <bool TA, bool TB, typename OutT>
void launch_gemm_t(const bf16* A, const bf16* B, OutT* C, const float* bias,
                   int M, int N, int K, int lda, int ldb, int ldc,
                   hipStream_t stream);

// N-way split-K NT GEMM: one launch, grid.z = nz (returned), partials in
// C (with bias) and nz-1 stacked slices of C2 (without); K must be a
// BK(=64) multiple with >= 2 tiles.
template <typename OutT>
int launch_gemm_splitk_t(const bf16* A, const bf16* B, OutT* C, OutT* C2,
                         const float* bias, int M, int N, int K, int lda,
                         int ldb, int ldc, hipStream_t stream);

// lstm.hip (fragment-packed operands; see lstm.hip header comment)
void launch_pack_gated_w(const bf16* W, bf16* out, int rows, int ngates,
                         int K, hipStream_t stream);
void launch_pack_a(const bf16* A, bf16* out, int B, int K,
                   hipStream_t stream);
void launch_lstm_cell_fwd(const bf16* h_pack, const float* c_prev,
                          const bf16* gx, const bf16* W_pack, bf16* h_out,
                          bf16* h_pack_out, float* c_out, bf16* gates_out,
                          bf16* rec, int B, int H, int HSp,
                          hipStream_t stream);
void launch_lstm_cell_bwd_elt(const bf16* dy, const float* dh_rec,
                              const float* dh_rec2, float* dc,
                              const bf16* rec, bf16* dG, bf16* dG_pack,
                              int B, int H, int HSp, hipStream_t stream);
void launch_smallm_packed_nt(const bf16* A_pack, const bf16* W_pack,
                             float* C, float* C2, int M, int N, int K,
                             hipStream_t stream);
void launch_smallm_fused_bwd2(const bf16* A_pack, const bf16* W_pack,
                              float* P, const bf16* dY, float* dc,
                              const bf16* rec, bf16* dG, bf16* pack_mid,
                              bf16* pack_out, unsigned int* flags,
                              unsigned int* pstate,
                              unsigned int* abort_flag, int M, int N, int K,
                              int HSp, int t, unsigned int step1,
                              unsigned int gen, int64_t rstep, int nthreads,
                              hipStream_t stream);
void launch_smallm_fused_bwd(const bf16* A_pack, const bf16* W_pack,
                             float* P, const bf16* dy, float* dc,
                             const bf16* rec, bf16* dG, bf16* dG_pack_out,
                             unsigned int* flags, unsigned int* abort_flag,
                             int M, int N, int K, int HSp, unsigned int step,
                             int nsplit, int nthreads, hipStream_t stream);

// lstm_persistent.hip — one launch for a whole layer unroll
int persistent_hs(int H);
void set_fwd_threads_impl(int v);
size_t persistent_fwd_lds(int B, int H);
void launch_lstm_persistent_fwd(const bf16* gx, const bf16* W_h, bf16* h_all,
                                float* c_all, bf16* rec,
                                unsigned int* pstate,
                                unsigned int* abort_flag, int T, int B,
                                int H, hipStream_t stream);
size_t persistent_bwd_lds(int B, int H);
void launch_lstm_persistent_bwd(const bf16* dY, const bf16* rec,
                                const bf16* W_h_T, bf16* dG, bf16* dG_packT,
                                unsigned int* pstate,
                                unsigned int* abort_flag, int T, int B,
                                int H, hipStream_t stream);

// elementwise.hip
void launch_embedding_fwd(const bf16* W, const int64_t* idx, bf16* out,
                          int N, int H, hipStream_t stream);
void launch_embedding_bwd_det(const bf16* dY, const int64_t* idx, float* dW,
                              int N, int H, int V, hipStream_t stream);
void launch_embedding_bwd(const bf16* dY, const int64_t* idx, float* dW,
                          int N, int H, hipStream_t stream);
void launch_dropout_fwd(const bf16* x, bf16* y, float p, uint64_t seed,
                        uint64_t offset, int64_t n, hipStream_t stream);
void launch_dropout_bwd(const bf16* dy, bf16* dx, float p, uint64_t seed,
                        uint64_t offset, int64_t n, hipStream_t stream);
void launch_lsm_nll_fwd(const float* scores, const int64_t* y, float* lse,
                        float* loss_accum, int N, int V, hipStream_t stream);
void launch_lsm_nll_bwd(const float* scores, const float* lse,
                        const int64_t* y, const float* upstream, float scale,
                        float* dscores, int N, int V, hipStream_t stream);
void launch_norm2_accum(const float* g, int64_t n, float* accum,
                        hipStream_t stream);
void launch_norm2_mt(const int64_t* desc, int nchunk, float* accum,
                     hipStream_t stream);
void launch_sgd_mt(const int64_t* desc, int nchunk, const float* norm2,
                   float max_norm, float lr, float grad_scale,
                   hipStream_t stream);
void launch_sgd_update(float* master, const float* grad, bf16* shadow,
                       const float* norm2, float max_norm, float lr,
                       float grad_scale, int64_t n, hipStream_t stream);
void launch_transpose_bf16(const bf16* src, bf16* dst, int R, int C,
                           int ldd, hipStream_t stream);
void launch_add2_f32_bf16(const float* a, const float* b, bf16* out,
                          int64_t n, hipStream_t stream);
void launch_addn_f32_bf16(const float* a, const float* extra, int nextra,
                          bf16* out, int64_t n, hipStream_t stream);
void launch_colsum_bf16(const bf16* in, float* out, int R, int C,
                        hipStream_t stream);
void launch_colsum_f32(const float* in, float* out, int R, int C,
                       hipStream_t stream);
void launch_softmax_acc(const float* scores, float* acc, int N, int V,
                        hipStream_t stream);

}  // namespace zamd
