/* dolomite_hip — C-ABI boundary of the MI355X HIP kernel library.
 *
 * These entry points are the gfx950-native replacements for the GPU work the
 * reference delegates to third-party kernels (SURVEY.md §2 "Delegated
 * kernel" table). Each declaration cites the reference call site
 * (paths under /root/reference/dolomite_engine/) it replaces.
 *
 * Conventions:
 *   - caller (the Python host on PyTorch-ROCm) owns all memory; pointers are
 *     device pointers; strides/sizes are in ELEMENTS (not bytes) as int64;
 *   - `stream` is the hipStream_t the call is ordered on (passed as void*);
 *   - dtype codes: 0 = fp32, 1 = bf16;
 *   - return 0 on success, nonzero hipError_t-style code on failure;
 *   - thread-safe per stream; no internal allocation on the hot path.
 */

#ifndef DOLOMITE_HIP_H
#define DOLOMITE_HIP_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef void* dolomite_stream_t; /* hipStream_t */

enum dolomite_dtype { DOLOMITE_F32 = 0, DOLOMITE_BF16 = 1 };

/* Library/version probe (used by the loader to verify ABI). */
int dolomite_hip_abi_version(void);

/* ------------------------------------------------------------------------
 * Fused RMSNorm (+ optional residual add), forward / backward.
 * Replaces: Triton RMSNorm (modeling_utils/normalization/rmsnorm/torchtitan.py)
 * with the canonical semantics of rmsnorm/base.py:18-25:
 *   s     = x + res_in            (when res_in != NULL; else s = x)
 *   rstd  = rsqrt(mean(s_fp32^2) + eps)
 *   y     = w * cast_to_input_dtype(s_fp32 * rstd)
 * res_out (when non-NULL) receives s (the pre-norm sum, used as the next
 * residual). rstd is cached fp32 per row for backward.
 * x,y,res_*: (T, H) row-major contiguous, dtype `dtype`. w: (H,), dtype.
 * ---------------------------------------------------------------------- */
int dolomite_rmsnorm_fwd(dolomite_stream_t stream,
                         const void* x, const void* res_in, const void* w,
                         void* y, void* res_out, float* rstd,
                         int64_t T, int64_t H, float eps, int dtype);

/* Backward: given dy and the saved pre-norm input s (= res_out of fwd, or x
 * when no residual) and rstd, computes
 *   dx  = r*(w*dy - s_hat * mean(w*dy*s_hat)) [+ dres_in]  (s_hat = s_fp32*rstd;
 *         dres_in, when non-NULL, is the residual-stream gradient folded into
 *         dx in the same pass instead of a separate elementwise add)
 *   dw  = sum_rows(dy * cast_to_dtype(s_hat))        (fp32 accumulation)
 * dw_partial: (nblocks, H) fp32 scratch written by the kernel;
 * dolomite_reduce_partials sums it into dw (H,) fp32.
 * nblocks is returned by dolomite_rmsnorm_bwd_nblocks(T). */
int dolomite_rmsnorm_bwd_nblocks(int64_t T);
int dolomite_rmsnorm_bwd(dolomite_stream_t stream,
                         const void* dy, const void* s, const void* w,
                         const float* rstd, void* dx, float* dw_partial,
                         const void* dres_in, int64_t T, int64_t H, int dtype);

/* LayerNorm, same calling pattern (replaces the reference 'layernorm'/'torch'
 * nn.LayerNorm path; mean/rstd cached fp32). db shares dw_partial layout:
 * dw_partial holds (2*nblocks, H): [dw partials ; db partials]. */
int dolomite_layernorm_fwd(dolomite_stream_t stream,
                           const void* x, const void* res_in, const void* w, const void* b,
                           void* y, void* res_out, float* mean, float* rstd,
                           int64_t T, int64_t H, float eps, int dtype);
int dolomite_layernorm_bwd(dolomite_stream_t stream,
                           const void* dy, const void* s, const void* w,
                           const float* mean, const float* rstd,
                           void* dx, float* dwdb_partial,
                           const void* dres_in, int64_t T, int64_t H, int dtype);

/* Sum partials: out[h] = sum_i partial[i*H + h]; out fp32 (H,). */
int dolomite_reduce_partials(dolomite_stream_t stream,
                             const float* partial, float* out,
                             int64_t nblocks, int64_t H);

/* ------------------------------------------------------------------------
 * RoPE on the packed QKV projection output, forward / backward.
 * Replaces apply_rotary_pos_emb (position_embedding/rope.py:104-121) at the
 * call site attention/padding_free.py:38-40, operating IN PLACE of the
 * fused c_attn layout (attention/base.py:72-81) without unpacking:
 *   qkv: (T, row_len) where row q-head h lives at
 *        (h/G)*q_gstride + (h%G)*D  and kv-head j at k_off + j*kv_hstride
 *        (v is untouched). G = n_head/num_kv_heads heads per kv group.
 *   cos/sin: (T, D) fp32, already gathered per token position
 *        (gpt_dolomite/base.py:289-296).
 *   dir: +1 forward rotation, -1 inverse (exact backward, since the table
 *        halves are duplicated: R^T = -R and S commutes with R).
 * out != in allowed (copy-through), in == out allowed (in-place).
 * ---------------------------------------------------------------------- */
int dolomite_rope_qkv(dolomite_stream_t stream,
                      const void* qkv_in, void* qkv_out,
                      const float* cos_t, const float* sin_t,
                      int64_t T, int64_t row_len,
                      int H, int Hkv, int D, int G,
                      int64_t q_gstride, int64_t k_off, int64_t kv_hstride,
                      int dir, int rotate_v_copy, int dtype);

/* ------------------------------------------------------------------------
 * Varlen causal flash attention, forward.
 * Replaces flash_attn_varlen_func (attention/padding_free.py:51-62).
 * q/k/v address layout (strides in elements):
 *   q[t, h, d]  = q_ptr + t*q_tstride + (h/G)*q_gstride + (h%G)*D + d
 *   k[t, j, d]  = k_ptr + t*k_tstride + j*k_hstride + d
 *   v[t, j, d]  = v_ptr + t*v_tstride + j*v_hstride + d
 * (this addresses the packed c_attn output without copies for MHA/GQA/MQA)
 * o: (T, H*D) row-major contiguous bf16/fp32; lse: (H, T) fp32 log-sum-exp
 * (natural log, includes softmax scale), for backward.
 * cu_seqlens: (batch+1,) int32 device pointer; causal only; dropout
 * unsupported (training configs on this path run attn_pdrop = 0).
 * ---------------------------------------------------------------------- */
int dolomite_fa_varlen_fwd(dolomite_stream_t stream,
                           const void* q, const void* k, const void* v,
                           void* o, float* lse,
                           const int32_t* cu_seqlens, int batch, int max_seqlen, int64_t T,
                           int H, int Hkv, int D, int G,
                           int64_t q_tstride, int64_t q_gstride,
                           int64_t k_tstride, int64_t k_hstride,
                           int64_t v_tstride, int64_t v_hstride,
                           float scale, int dtype);

/* Backward, three passes (all launched by the host in order on `stream`):
 *  1. preprocess: delta[h,t] = rowsum(dO[t,h,:]*O[t,h,:]) fp32.
 *  2. main: dkv — one workgroup per (kv-tile, seq, q-head); recomputes P
 *     from q,k,lse; dk/dv accumulate in registers over the q-tile loop and
 *     store EXCLUSIVELY (no atomics) into (T, H, D) fp32 per-q-head
 *     partials. dq — one workgroup per (q-tile, seq, q-head),
 *     register-accumulated and stored bf16 directly into the packed dqkv
 *     q-slots.
 *  3. grad_finalize: reduces each kv head's G partial contributions in
 *     FIXED order (deterministic gradients — bit-exact resume depends on
 *     it) and casts into the packed dqkv.
 */
int dolomite_fa_bwd_preprocess(dolomite_stream_t stream,
                               const void* o, const void* dout, float* delta,
                               int64_t T, int H, int D,
                               int64_t o_tstride, int64_t do_tstride, int dtype);
int dolomite_fa_varlen_bwd(dolomite_stream_t stream,
                           const void* q, const void* k, const void* v,
                           const void* dout, const float* lse,
                           const float* delta, void* dqkv_q, float* dk_acc, float* dv_acc,
                           const int32_t* cu_seqlens, int batch, int max_seqlen, int64_t T,
                           int H, int Hkv, int D, int G,
                           int64_t q_tstride, int64_t q_gstride,
                           int64_t k_tstride, int64_t k_hstride,
                           int64_t v_tstride, int64_t v_hstride,
                           int64_t do_tstride,
                           float scale, int dtype);
/* dk_acc/dv_acc are (T, H, D) fp32 PER-Q-HEAD partials written exclusively
 * (no atomics) by dolomite_fa_varlen_bwd; the finalize reduces each kv
 * head's G contributors in fixed order — deterministic gradients. */
int dolomite_fa_grad_finalize(dolomite_stream_t stream,
                              const float* dk_acc, const float* dv_acc,
                              void* dqkv, int64_t T, int Hkv, int D, int G,
                              int64_t row_tstride,
                              int64_t k_off, int64_t kv_hstride, int64_t v_off, int dtype);

/* MFMA fragment-layout self-test: computes C = A(16x32) @ B(32x16) with the
 * compiled fragment mapping; host verifies against a CPU matmul.
 * A/B bf16 row-major contiguous, C fp32 row-major. */
int dolomite_mfma_probe(dolomite_stream_t stream,
                        const void* A, const void* B, float* C);

/* ------------------------------------------------------------------------
 * Fused softmax cross-entropy over the vocab, forward / backward.
 * Replaces F.cross_entropy at model_wrapper/pretraining.py:125 and
 * gpt_dolomite/main.py:200 (mean reduction, ignore_index -100, fp32
 * log-softmax as torch does for bf16 inputs).
 * fwd: per-row loss (T,) fp32 (0 for ignored rows), per-row lse (T,) fp32,
 *      n_valid is computed by the caller.
 * bwd: dlogits = (softmax - onehot) * grad_scale, 0 for ignored rows;
 *      dlogits may alias logits (in-place).
 * logits rows at logits_ptr + t*row_stride.
 * ---------------------------------------------------------------------- */
int dolomite_ce_fwd(dolomite_stream_t stream,
                    const void* logits, const int64_t* labels,
                    float* row_loss, float* lse,
                    int64_t T, int64_t V, int64_t row_stride,
                    int ignore_index, int dtype);
/* grad_scale_dev: DEVICE pointer to one fp32 (upstream grad / n_valid),
 * so the backward launch needs no host readback of device scalars. */
int dolomite_ce_bwd(dolomite_stream_t stream,
                    const void* logits, const int64_t* labels, const float* lse,
                    void* dlogits, const float* grad_scale_dev,
                    int64_t T, int64_t V, int64_t row_stride,
                    int ignore_index, int dtype);

/* ------------------------------------------------------------------------
 * Fused AdamW on a flat fp32 master shard, with bf16 param write-out.
 * Replaces the TorchAdamW step (optimization/optimizer.py:74, defaults
 * arguments.py:235-249) on the ZeRO-2 local shard. torch.optim.AdamW
 * semantics: decoupled wd, bias-corrected moments.
 *   master/m/v: (n,) fp32.  grad: (n,) fp32 or bf16 (grad_dtype).
 *   param_out: (n,) bf16 copy of updated master (NULL to skip).
 *   gscale: optional DEVICE scalar multiplied into every grad element
 *           (the grad-clip coefficient, fused here so the clip costs one
 *           scalar read instead of a read+write pass; NULL = 1.0).
 * ---------------------------------------------------------------------- */
int dolomite_adamw_step(dolomite_stream_t stream,
                        float* master, void* param_out_bf16,
                        const void* grad, int grad_dtype,
                        float* m, float* v,
                        int64_t n, float lr, float beta1, float beta2,
                        float eps, float weight_decay, int step,
                        const float* gscale);

/* Multiply a flat fp32 (or bf16) buffer by a scalar (grad clip apply). */
int dolomite_scale_inplace(dolomite_stream_t stream, void* buf, int64_t n,
                           float scale, int dtype);

/* Deterministic MoE top-k row combine: out[t,:] = sum_j h[inv[t*k_top+j],:]
 * (fixed j order). Replaces torch zeros+index_add for the expert scatter-back
 * (moe/base.py:127) and the backward of the expert-row gather — both ~9x off
 * the HBM roofline in eager torch. inv: (T*k_top,) int32 inverse of the
 * expert sort (inv[pair] = slot). bf16, K % 8 == 0. */
int dolomite_moe_rows_combine(dolomite_stream_t stream, const void* h,
                              const int32_t* inv, void* out,
                              int64_t T, int K, int k_top, int dtype);

/* Deterministic sum of squares (grad-norm input): out[0] = sum(x[i]^2) in
 * fp32, fixed reduction order run-to-run (per-thread strided order + fixed
 * LDS trees), so the clipped optimizer step stays bit-reproducible.
 * partials: (1024,) fp32 scratch. x: (n,) fp32 or bf16. */
int dolomite_sqsum(dolomite_stream_t stream, const void* x, int64_t n,
                   float* partials, float* out, int dtype);

/* ------------------------------------------------------------------------
 * Grouped expert GEMMs for the MoE family — replaces the reference's
 * scattermoe/triton grouped path (moe_dolomite/moe/scatter.py:109-138) for
 * the SparseMoE expert matmuls (moe/base.py:137-156). Rows are
 * expert-sorted; offsets is the (E+1) int32 cumulative row count per
 * expert (device pointer). bf16 in / fp32 accumulate / bf16 out.
 * Requires K % 8 == 0 and N % 8 == 0 (error 9020 otherwise; the Python
 * layer falls back to the eager per-expert loop).
 *   fwd   : y[t,n]    = sum_k x[t,k] * w[e(t),n,k] (+ bias[e,n], opt NULL)
 *   dgrad : dx[t,k]   = sum_n dy[t,n] * w[e(t),n,k]
 *   wgrad : dw[e,n,k] = sum_{t in group e} dy[t,n] * x[t,k]
 * max_rows = largest group size (host int, for grid sizing).
 * ---------------------------------------------------------------------- */
int dolomite_moe_gemm_fwd(dolomite_stream_t stream, const void* x, const void* w,
                          const void* bias, void* y, const int32_t* offsets,
                          int E, int max_rows, int N, int K, int dtype);
int dolomite_moe_gemm_dgrad(dolomite_stream_t stream, const void* dy, const void* w,
                            void* dx, const int32_t* offsets,
                            int E, int max_rows, int N, int K, int dtype);
int dolomite_moe_gemm_wgrad(dolomite_stream_t stream, const void* dy, const void* x,
                            void* dw, const int32_t* offsets,
                            int E, int N, int K, int dtype);

/* ------------------------------------------------------------------------
 * Host-side dataset index builders (CPU; the reference's only native
 * component, data/megatron/utils/helpers.cpp).
 * build_sample_idx: pack epoch-replicated documents into seq_length+1 token
 *   windows; sample_idx is [(num_samples+1) x 2] rows of (doc_idx position,
 *   token offset); caller computes num_samples =
 *   (num_epochs*tokens_per_epoch - 1) / seq_length and allocates.
 * build_blending_indices: greedy max-error dataset interleave. Buffers are
 * HOST pointers.
 * ---------------------------------------------------------------------- */
int dolomite_build_sample_idx_i32(const int32_t* sizes, const int32_t* doc_idx,
                                  int32_t seq_length, int32_t num_epochs,
                                  int64_t tokens_per_epoch, int32_t* sample_idx,
                                  int64_t num_samples);
int dolomite_build_sample_idx_i64(const int32_t* sizes, const int32_t* doc_idx,
                                  int32_t seq_length, int32_t num_epochs,
                                  int64_t tokens_per_epoch, int64_t* sample_idx,
                                  int64_t num_samples);
int dolomite_build_blending_indices(int16_t* dataset_index, int64_t* dataset_sample_index,
                                    const double* weights, int32_t num_datasets,
                                    int64_t size);

#ifdef __cplusplus
}
#endif

#endif /* DOLOMITE_HIP_H */
