// C ABI between the HIP kernel library (hipcc-compiled, torch-free) and the
// torch extension binding (csrc/bind/kernels_binding.cpp).
// dtype codes: 0 = f32, 1 = bf16, 2 = f16.  All activation tensors are NHWC
// physical (torch channels_last); M = N*H*W rows over C channels.
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

extern "C" {

// ---- elementwise / utility ------------------------------------------------
void ddl_relu_bwd(const void* dy, const void* y, void* out, int64_t n,
                  int dtype, hipStream_t s);
void ddl_check_finite(const void* g, int64_t n, int dtype, int* found_inf,
                      hipStream_t s);
void ddl_unscale_check(void* g, int64_t n, int dtype, float inv_scale,
                       int* found_inf, hipStream_t s);
void ddl_dropout_fwd(const void* x, void* y, uint8_t* mask, int64_t n,
                     int dtype, float p, uint64_t seed, hipStream_t s);
void ddl_dropout_bwd(const void* dy, void* dx, const uint8_t* mask, int64_t n,
                     int dtype, float p, hipStream_t s);
void ddl_cast_f32_to(const float* src, void* dst, int64_t n, int dtype,
                     hipStream_t s);

// ---- fused optimizer (K10/K11) -------------------------------------------
// table: rows of [param_ptr, grad_ptr, master_ptr, mom_ptr, numel, use_lars,
// tensor_idx] (int64), one row per block.  norms: [2, ntensors] f32
// (0: sum w^2 of master, 1: sum g'^2 where g' = g*inv_scale + wd*w).
void ddl_fused_sgd(const int64_t* table, int nrows, const float* norms,
                   int ntensors, int dtype, float lr, float momentum,
                   float wd, int nesterov, float lars_eta, float lars_eps,
                   float inv_scale, hipStream_t s);
// deterministic two-stage: per-chunk partials[2*nrows] then a per-tensor
// fold in fixed chunk order (no atomics)
void ddl_lars_norms(const int64_t* table, int nrows, float* norms,
                    float* partials, int ntensors, int dtype, float wd,
                    float inv_scale, hipStream_t s);

// ---- fused BatchNorm + residual + ReLU (K4/K5/K14) ------------------------
// sums: f32 [2*C]; partials: f32 [chunks*2*C] scratch where chunks =
// ddl_bn_stats_chunks(M, C).  Two-stage deterministic reduction.
int ddl_bn_stats_chunks(int64_t M, int C);
void ddl_bn_stats(const void* x, int64_t M, int C, int dtype, float* sums,
                  float* partials, int max_chunks, hipStream_t s);
// finalize: mean/rstd from sums; optionally update running stats in-place.
// fold conv-epilogue partials [chunks][2][C] -> sums[2][C] (deterministic,
// two stage for large chunk counts) and finalize mean/rstd.
void ddl_bn_reduce_finalize(const float* partials, float* fold_ws, int chunks,
                            int C, int64_t M, float momentum, float eps,
                            float* running_mean, float* running_var,
                            float* mean_out, float* rstd_out,
                            int update_running, hipStream_t s);
void ddl_bn_finalize(const float* sums, int C, int64_t M, float momentum,
                     float eps, float* running_mean, float* running_var,
                     float* mean_out, float* rstd_out, int update_running,
                     hipStream_t s);
// y = relu?( (x-mean)*rstd*w + b + residual? )   (w/b in x's dtype)
void ddl_bn_apply(const void* x, void* y, const void* residual,
                  const float* mean, const float* rstd, const void* weight,
                  const void* bias, int64_t M, int C, int relu, int dtype,
                  hipStream_t s);
// inference: scale/shift from running stats, single pass
void ddl_bn_infer(const void* x, void* y, const void* residual,
                  const float* running_mean, const float* running_var,
                  float eps, const void* weight, const void* bias, int64_t M,
                  int C, int relu, int dtype, hipStream_t s);
// backward stats: sums[0..C-1] = sum(dy_m * xhat), sums[C..] = sum(dy_m).
// relu_mode: 0 none; 1 mask from y (residual); 2 mask recomputed as
// (w*xhat+b > 0) — skips the y read (one activation pass saved).
void ddl_bn_bwd_stats(const void* dy, const void* x, const void* y,
                      int relu_mode, const float* mean, const float* rstd,
                      const void* weight, const void* bias, int64_t M, int C,
                      int dtype, float* sums, float* partials, int max_chunks,
                      hipStream_t s);
// dx (+ optional dres copy of masked dy), dweight/dbias cast from sums.
void ddl_bn_bwd_apply(const void* dy, const void* x, const void* y,
                      void* dx, void* dres, const float* mean,
                      const float* rstd, const void* weight, const void* bias,
                      const float* sums, int64_t M, int C, int relu_mode,
                      int dtype, hipStream_t s);
void ddl_bn_grads_from_sums(const float* sums, void* dweight, void* dbias,
                            int C, int dtype, hipStream_t s);
// mask-aware variants: relu_mask is the bit-packed [M][C/8] ReLU activity
// written by the forward apply (mode 3 backward reads it instead of y).
void ddl_bn_apply_mask(const void* x, void* y, const void* residual,
                       const float* mean, const float* rstd,
                       const void* weight, const void* bias, int64_t M,
                       int C, int relu, int dtype, uint8_t* relu_mask,
                       hipStream_t s);
void ddl_bn_bwd_stats_mask(const void* dy, const void* x, const void* y,
                           int relu_mode, const float* mean,
                           const float* rstd, const void* weight,
                           const void* bias, int64_t M, int C, int dtype,
                           float* sums, float* partials, int max_chunks,
                           const uint8_t* relu_mask, hipStream_t s);
void ddl_bn_bwd_apply_mask(const void* dy, const void* x, const void* y,
                           void* dx, void* dres, const float* mean,
                           const float* rstd, const void* weight,
                           const void* bias, const float* sums, int64_t M,
                           int C, int relu_mode, int dtype,
                           const uint8_t* relu_mask, hipStream_t s);

// ---- pooling (K6/K7) ------------------------------------------------------
void ddl_maxpool_fwd(const void* x, void* y, uint8_t* idx, int N, int H,
                     int W, int C, int P, int Q, int kh, int kw, int sh,
                     int sw, int ph, int pw, int dtype, hipStream_t s);
void ddl_maxpool_bwd(const void* dy, const uint8_t* idx, void* dx, int N,
                     int H, int W, int C, int P, int Q, int kh, int kw,
                     int sh, int sw, int ph, int pw, int dtype, hipStream_t s);
void ddl_gap_fwd(const void* x, void* y, int N, int HW, int C, int dtype,
                 hipStream_t s);
void ddl_gap_bwd(const void* dy, void* dx, int N, int HW, int C, int dtype,
                 hipStream_t s);

// ---- fused softmax + cross-entropy (K9) -----------------------------------
void ddl_softmax_xent_fwd(const void* logits, const int64_t* target,
                          float* losses, float* probs, int N, int C,
                          int dtype, float label_smoothing, hipStream_t s);
void ddl_softmax_xent_bwd(const float* probs, const int64_t* target,
                          void* dlogits, int N, int C, int dtype,
                          float label_smoothing, const float* scale_ptr,
                          float scale_const, hipStream_t s);

// ---- implicit-GEMM conv (K1/K2/K3), NHWC, MFMA ----------------------------
// x [N,H,W,C], w [K,R,S,C] (KRSC = channels_last), y [N,P,Q,K].
// conv forward that also emits BN {sum,sumsq} partials [cdiv(M,128)][2][K]
// from its epilogue (2-byte dtypes, vectorizable C; stats_ws may be null).
void ddl_conv_fwd_stats(const void* x, const void* w, const void* bias,
                        void* y, float* stats_ws, int N, int H, int W, int C,
                        int K, int R, int S, int P, int Q, int sh, int sw,
                        int ph, int pw, int relu, int dtype, hipStream_t s);
int64_t ddl_conv_fwd_stats_chunks(int N, int P, int Q, int C, int K, int R,
                                  int S, int dtype);
void ddl_conv_fwd(const void* x, const void* w, const void* bias, void* y,
                  int N, int H, int W, int C, int K, int R, int S, int P,
                  int Q, int sh, int sw, int ph, int pw, int relu, int dtype,
                  hipStream_t s);
// dgrad consumes w_t [C,R,S,K] (CRSK) built by ddl_transpose_krsc_crsk.
void ddl_conv_dgrad_accum(const void* dy, const void* w_t, void* dx,
                          const void* accum, int N, int H, int W, int C,
                          int K, int R, int S, int P, int Q, int sh, int sw,
                          int ph, int pw, int dtype, hipStream_t s);
void ddl_conv_dgrad(const void* dy, const void* w_t, void* dx, int N, int H,
                    int W, int C, int K, int R, int S, int P, int Q, int sh,
                    int sw, int ph, int pw, int dtype, hipStream_t s);
// dgrad that also emits the PRECEDING BatchNorm's backward partials
// [chunks][2][C] {sum(g*xhat), sum(g)} from its epilogue (2-byte dtypes;
// relu_mode-2 mask recomputed from the BN's input bnx).  That BN then
// skips its bn_bwd_stats pass.
int64_t ddl_conv_dgrad_bnstats_chunks(int N, int H, int W, int C, int K,
                                      int sh, int sw);
// accum: optional residual-join gradient added in the epilogue BEFORE the
// stats (the fork-conv case); bn_mask: optional bit-packed ReLU mask from
// the BN's forward (residual-join BNs) replacing the recomputed mask.
void ddl_conv_dgrad_bnstats(const void* dy, const void* w_t, void* dx,
                            const void* bnx, const float* bn_mean,
                            const float* bn_rstd, const void* bn_w,
                            const void* bn_b, float* bstats_ws,
                            const void* accum, const uint8_t* bn_mask,
                            int N, int H, int W, int C, int K, int R, int S,
                            int P, int Q, int sh, int sw, int ph, int pw,
                            int dtype, hipStream_t s);
// reduce bwd partials [chunks][2][C] -> sums[2][C] (deterministic)
void ddl_bn_reduce_sums(const float* partials, float* fold_ws, int chunks,
                        int C, hipStream_t s);
// wgrad: fp32 per-chunk slabs in dw_ws (ddl_conv_wgrad_ws_elems floats,
// no pre-zero needed) reduced+cast into dw (KRSC, dtype).  Deterministic.
int64_t ddl_conv_wgrad_ws_elems(int N, int C, int K, int R, int S, int P,
                                int Q);
void ddl_conv_wgrad(const void* dy, const void* x, float* dw_ws, void* dw,
                    int N, int H, int W, int C, int K, int R, int S, int P,
                    int Q, int sh, int sw, int ph, int pw, int dtype,
                    hipStream_t s);
// im2col (small-C stem): x -> [N*P*Q, ldout] with (r,s,c) flattened cols
// zero-padded to ldout (multiple of 8); plus K x kred <-> K x ldout pads.
void ddl_im2col(const void* x, void* out, const int* tbl, int N, int H,
                int W, int C, int R, int S, int P, int Q, int sh, int sw,
                int ph, int pw, int ldout, int dtype, hipStream_t s);
// col2im scatter-add for the stem dgrad (fp32 atomic accumulation)
void ddl_col2im_acc(const void* dcol, float* dx_ws, const int* tbl, int N,
                    int H, int W, int C, int P, int Q, int sh, int sw,
                    int ph, int pw, int kred_total, int ldout, int dtype,
                    hipStream_t s);
void ddl_pad_cols(const void* w, void* out, int K, int kred, int ldout,
                  int dtype, hipStream_t s);
void ddl_narrow_cols(const void* w, void* out, int K, int kred, int ldout,
                     int dtype, hipStream_t s);
// [K, RS, C] -> [C, RS, K] element transpose (K12 utility).
void ddl_transpose_krsc_crsk(const void* w, void* w_t, int K, int RS, int C,
                             int dtype, hipStream_t s);

}  // extern "C"
