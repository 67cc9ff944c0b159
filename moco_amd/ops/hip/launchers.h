// C ABI between the torch binding TU (bindings.cpp) and the pure-HIP kernel
// TUs (*.hip).  `stream` is a hipStream_t passed as void*.
// Every launcher returns a hipError_t as int (0 == success).
#pragma once

#include <cstdint>

extern "C" {

// ---- l2norm.hip -----------------------------------------------------------
// x: [N, C] (bf16 if is_bf16 else fp32) -> y same dtype, inv_norm fp32 [N]
int launch_l2norm_fwd(void* stream, const void* x, void* y, float* inv_norm,
                      int64_t N, int64_t C, bool is_bf16);
int launch_l2norm_bwd(void* stream, const void* g, const void* y,
                      const float* inv_norm, void* gx, int64_t N, int64_t C,
                      bool is_bf16);

// ---- contrastive.hip ------------------------------------------------------
// q,k: [N,C] bf16; queue_kc: [K,C] bf16; out: [N,1+K] fp32
int launch_contrastive_fwd(void* stream, const void* q, const void* k,
                           const void* queue_kc, float* out, int64_t N,
                           int64_t C, int64_t K, float invT);
// g: [N,1+K] fp32; k: [N,C] bf16; queue_ck: [C,K] fp32; dq (pre-zeroed): [N,C] fp32
int launch_contrastive_bwd(void* stream, const float* g, const void* k,
                           const float* queue_ck, float* dq, int64_t N,
                           int64_t C, int64_t K, float invT);

// ---- softmax_ce.hip -------------------------------------------------------
// logits: [N, M] fp32
int launch_softmax_ce0_fwd(void* stream, const float* logits, float* loss_rows,
                           float* row_max, float* row_lse, int64_t N,
                           int64_t M);
int launch_softmax_ce0_bwd(void* stream, const float* logits,
                           const float* row_max, const float* row_lse,
                           float* dlogits, float scale, const float* gscale,
                           int64_t N, int64_t M);

// ---- queue.hip ------------------------------------------------------------
// queue: [C,K] fp32; queue_kc: [K,C] bf16 (nullable); keys: [B,C]
int launch_queue_enqueue(void* stream, float* queue, void* queue_kc,
                         const void* keys, bool keys_bf16, int64_t C,
                         int64_t K, int64_t B, int64_t ptr);
// graph-replay-safe variant: ring position read/advanced from ptr_dev
int launch_queue_enqueue_dev(void* stream, float* queue, void* queue_kc,
                             const void* keys, bool keys_bf16, int64_t C,
                             int64_t K, int64_t B, int64_t* ptr_dev);

// ---- multi_tensor.hip -----------------------------------------------------
int launch_multi_tensor_ema(void* stream, const int64_t* ptrs_dst,
                            const int64_t* ptrs_src, const int32_t* sizes,
                            int64_t nchunks, float m);
int launch_multi_tensor_sgd(void* stream, const int64_t* ptrs_p,
                            const int64_t* ptrs_g, const int64_t* ptrs_b,
                            const int32_t* sizes, int64_t nchunks, float lr,
                            float momentum, float wd);
int launch_ema_flat(void* stream, float* dst, const float* src, int64_t n,
                    float m);

// ---- bn.hip ---------------------------------------------------------------
// x/y/g/res: [R, C] bf16 (NHWC flattened); stats fp32; C in {64..256 dividing
// 256, or multiples of 256 up to 2048}; part_* workspaces are [304, C] fp32.
int launch_bn_stats(void* stream, const void* x, float* part_s, float* part_q,
                    float* mean, float* rstd, float* running_mean,
                    float* running_var, int64_t R, int64_t C, float eps,
                    float momentum);
int launch_bn_apply(void* stream, const void* x, const void* res, void* y,
                    const float* mean, const float* rstd, const float* gamma,
                    const float* beta, int64_t R, int64_t C, bool relu);
int launch_bn_bwd_reduce(void* stream, const void* g, const void* y,
                         const void* x, const float* mean, const float* rstd,
                         float* part_db, float* part_dg, float* dbeta,
                         float* dgamma, int64_t R, int64_t C, bool relu);
int launch_bn_bwd_dx(void* stream, const void* g, const void* y,
                     const void* x, const float* mean, const float* rstd,
                     const float* gamma, const float* dbeta,
                     const float* dgamma, void* dx, void* dres, int64_t R,
                     int64_t C, bool relu, bool train);
// finalize mean/rstd (+ running stats) from the [P, C] slot-striped
// (sum, sumsq) partials produced by the conv-GEMM stats epilogue
int launch_bn_finalize(void* stream, const float* ssum, const float* ssq,
                       float* mean, float* rstd, float* running_mean,
                       float* running_var, int64_t R, int64_t C, int64_t P,
                       float eps, float momentum, bool zero_src);
// reduce two [P, C] slot slabs to [C] totals, re-zeroing the slabs
int launch_slot_reduce2(void* stream, float* s1, float* s2, float* o1,
                        float* o2, int64_t C, int64_t P);

// ---- augment.hip ----------------------------------------------------------
// src: [B,3,H,W] fp32; params: [M,12] fp32; out: [M,3,S,S] (bf16 or fp32,
// NCHW or NHWC per flags); tmp1/tmp2: [M,3,S,S] fp32 workspaces;
// graysum: [M] fp32 (pre-zeroed)
int launch_two_crop_augment(void* stream, const void* src, bool src_u8,
                            const float* params, void* out, float* tmp1,
                            float* tmp2, float* graysum, int64_t B, int64_t M,
                            int64_t H, int64_t W, int64_t S, bool out_bf16,
                            bool channels_last);

}  // extern "C"

// ---- pool.hip -------------------------------------------------------------
#ifdef __cplusplus
extern "C" {
#endif
// 3x3 stride-2 pad-1 max pool, NHWC bf16; idx: per-output uint8 tap id [0,9)
int launch_maxpool_fwd(void* stream, const void* x, void* y, void* idx,
                       int64_t N, int64_t C, int64_t H, int64_t W, int64_t Ho,
                       int64_t Wo);
int launch_maxpool_bwd(void* stream, const void* g, const void* idx, void* dx,
                       int64_t N, int64_t C, int64_t H, int64_t W, int64_t Ho,
                       int64_t Wo);
#ifdef __cplusplus
}
#endif

// ---- conv1x1.hip ----------------------------------------------------------
#ifdef __cplusplus
extern "C"
#endif
// C[M,N] = A[M,K] @ Bt[N,K]^T, bf16 in/out, fp32 accumulate (1x1-conv GEMM)
int launch_gemm_nt_bf16(void* stream, const void* A, const void* Bt, void* C,
                        int64_t M, int64_t N, int64_t K);
#ifdef __cplusplus
extern "C"
#endif
// c_add (nullable): epilogue C += c_add (requires the LDS-conforming shape)
int launch_gemm_nt_lds_bf16(void* stream, const void* A, const void* Bt,
                            void* C, int64_t M, int64_t N, int64_t K,
                            const void* c_add);
#ifdef __cplusplus
extern "C"
#endif
// a_out (nullable, needs a_scale): materialized transformed A operand [M,K]
int launch_gemm_nt_fused_bf16(void* stream, const void* A, const void* Bt,
                              void* C, int64_t M, int64_t N, int64_t K,
                              const float* a_scale, const float* a_shift,
                              bool a_relu, float* stats_sum,
                              float* stats_sumsq, void* a_out,
                              const void* c_add);
#ifdef __cplusplus
extern "C"
#endif
// 3x3 pad-1 conv (stride 1/2) as implicit MFMA GEMM (round-2 prototype).
// zero_page: [Ci + 8] bf16 — zeros, or the BN-apply neutral values.
int launch_conv3x3_fwd(void* stream, const void* x, const void* w,
                       const void* zero_page, void* out, int64_t N, int64_t H,
                       int64_t W, int64_t Ci, int64_t Co, int64_t stride,
                       const float* a_scale, const float* a_shift,
                       bool a_relu, float* stats_sum, float* stats_sumsq);
#ifdef __cplusplus
extern "C"
#endif
// strided 1x1 conv (downsample path), NHWC bf16 (round-2 prototype)
int launch_conv1x1_strided(void* stream, const void* x, const void* w,
                           void* out, int64_t N, int64_t H, int64_t W,
                           int64_t Ci, int64_t Co, int64_t stride);
#ifdef __cplusplus
extern "C"
#endif
// dgrad GEMM + fused BN-backward epilogue (round-2 prototype): C = relu-
// masked A@Bt^T, dbeta/dgamma partials accumulated (pre-zeroed, fp32).
int launch_gemm_nt_dgrad_epi(void* stream, const void* A, const void* Bt,
                             void* C, int64_t M, int64_t N, int64_t K,
                             const void* y, const void* z, const float* mean,
                             const float* rstd, float* dbeta, float* dgamma,
                             const void* c_add);

#ifdef __cplusplus
extern "C" {
#endif
// ---- weight mirrors (multi_tensor.hip) ------------------------------------
int launch_multi_cast_bf16(void* stream, const int64_t* src_ptrs,
                           const int64_t* dst_ptrs, const int32_t* sizes,
                           int64_t nchunks);
int launch_multi_transpose_bf16(void* stream, const int64_t* src_ptrs,
                                const int64_t* dst_ptrs, const int32_t* meta,
                                int64_t ntiles);
#ifdef __cplusplus
}
#endif

#ifdef __cplusplus
extern "C"
#endif
// fp8 e4m3 GEMM on the MX-scaled MFMA (per-tensor software scaling):
// C_bf16[M,N] = (A8[M,K] @ B8t[N,K]^T) * out_scale; optional bn-stats
// epilogue into [256, N] slot slabs.  M%128, N%64, K%128.
int launch_gemm_nt_fp8(void* stream, const void* A, const void* Bt, void* C,
                       int64_t M, int64_t N, int64_t K, float out_scale,
                       float* stats_sum, float* stats_sumsq);

#ifdef __cplusplus
extern "C" {
#endif
// head adaptive-avg-pool to 1x1, NHWC bf16
int launch_avgpool_fwd(void* stream, const void* x, void* y, int64_t N,
                       int64_t C, int64_t HW);
int launch_avgpool_bwd(void* stream, const void* g, void* dx, int64_t N,
                       int64_t C, int64_t HW);
#ifdef __cplusplus
}
#endif
