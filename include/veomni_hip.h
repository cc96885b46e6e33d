/* veomni_hip.h — C ABI of libveomni_hip.so (gfx950 / MI355X hot-path kernels).
 *
 * This is the drop-in boundary under the Python operator API.
 * Each entry replaces one reference kernel (anchor cited per function).
 * Conventions:
 *   - caller owns every buffer (device pointers from torch tensors);
 *   - all entries are stream-ordered on the passed hipStream_t (as void*);
 *   - return 0 on success, nonzero error code otherwise; vh_last_error()
 *     returns a thread-local message for the last failure;
 *   - no threads are spawned inside; no allocation except where stated;
 *   - bf16 buffers are passed as uint16_t*.
 */
#ifndef VEOMNI_HIP_H
#define VEOMNI_HIP_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Last error message (thread-local). */
const char* vh_last_error(void);

/* Build tag: returns a static string identifying arch + version. */
const char* vh_build_info(void);

/* ---- MoE token bookkeeping --------------------------------------------- */

/* Histogram of expert ids. Bit-exact.
 * Replaces: veomni/ops/kernels/moe/_kernels/kernel/moe.py:29-82.
 * expert_index: int64 [n]; out: int32 [num_experts] (pre-zeroed NOT required;
 * kernel zeroes it). */
int vh_expert_histogram(const int64_t* expert_index, int64_t n,
                        int num_experts, int32_t* out, void* stream);

/* Row scatter: out[index[i]] = x[i / topk]'s row — i.e. for each token row t
 * and slot k, out[index[t*topk+k]] = x[t]. bf16 rows.
 * Replaces: kernel/moe.py:253-333 (_moe_scatter_kernel).
 * x: [M, N] bf16; index: int32 [M*topk]; out: [M*topk, N] bf16. */
int vh_moe_scatter_bf16(const uint16_t* x, const int32_t* index, uint16_t* out,
                        int64_t M, int64_t N, int topk, void* stream);

/* Row gather with fp32 accumulation over topk:
 * out[t] = sum_k x[index[t*topk+k]].
 * Replaces: kernel/moe.py:87-159 (_moe_gather_kernel).
 * x: [M*topk, N] bf16; out: [M, N] bf16. */
int vh_moe_gather_bf16(const uint16_t* x, const int32_t* index, uint16_t* out,
                       int64_t M, int64_t N, int topk, void* stream);

/* ---- Grouped GEMM (MFMA bf16, fp32 accumulate) ------------------------- */

/* Per-group GEMM, shared N,K; group g owns rows [cumsum[g-1], cumsum[g]) of
 * A and C.  trans_b!=0: B is [G,N,K] and C_g = A_g @ B_g^T;
 * trans_b==0: B is [G,K,N] and C_g = A_g @ B_g.
 * activation: 0 none; 1 fused SiLU on the output (bf16 rounding after).
 * accumulate!=0: C += result (C read as bf16).
 * cumsum: int64 device [G], inclusive.
 * Replaces: kernel/group_gemm.py:66-234 (group_gemm_same_nk_kernel).
 * Constraints (v1): N % 16 == 0, K % 64 == 0. */
int vh_group_gemm_nk_bf16(const uint16_t* A, const uint16_t* B, uint16_t* C,
                          const int64_t* cumsum, int G, int64_t N, int64_t K,
                          int64_t total_rows, int trans_b, int accumulate,
                          int activation, void* stream);

/* Deep-pipelined 256x256 variant of vh_group_gemm_nk_bf16 (same semantics,
 * no accumulate/activation); auto-dispatched for large shapes, exported for
 * direct benchmarking. K % 32 == 0, K >= 64. */
int vh_group_gemm_nk8_bf16(const uint16_t* A, const uint16_t* B, uint16_t* C,
                           const int64_t* cumsum, int G, int64_t N, int64_t K,
                           int64_t total_rows, int trans_b, void* stream);

/* 256-square double-buffered glds variant (trans_b semantics; K % 64 == 0);
 * kept for A/B benchmarking — the dispatched fast path is nk256s below. */
int vh_group_gemm_nk256_bf16(const uint16_t* A, const uint16_t* B, uint16_t* C,
                             const int64_t* cumsum, int G, int64_t N,
                             int64_t K, int64_t total_rows, void* stream);

/* nk256 inner loop under a device-built tile schedule with XCD-clustered
 * persistent blocks (L2 reuse of the per-group A/B tiles; no skew-sized
 * null-block grid). The dispatched trans_b fast path. G <= 4096. NOTE: on
 * first use this entry hipMallocs one small (~16 KB) schedule workspace
 * that lives for the process (the single exception to the no-allocation
 * convention; calls on one stream serialize on it). */
int vh_group_gemm_nk256s_bf16(const uint16_t* A, const uint16_t* B,
                              uint16_t* C, const int64_t* cumsum, int G,
                              int64_t N, int64_t K, int64_t total_rows,
                              void* stream);

/* probe: nk256s with the 32x32x16 MFMA inner loop (issue-stall attack). */
int vh_group_gemm_nk256s32_bf16(const uint16_t* A, const uint16_t* B,
                                uint16_t* C, const int64_t* cumsum, int G,
                                int64_t N, int64_t K, int64_t total_rows,
                                void* stream);

/* Probe kernels kept for on-box A/B (trans_b semantics, XCD schedule):
 * nk8s = KSUB=32 counted-vmcnt 4-deep ring; nkp = 8-phase K-split pipeline
 * reconstruction. Both race-screened correct; neither dispatched (measured
 * slower than nk256s — see profiles/r02_groupgemm_pmc.txt / DESIGN.md). */
int vh_group_gemm_nk8s_bf16(const uint16_t* A, const uint16_t* B, uint16_t* C,
                            const int64_t* cumsum, int G, int64_t N,
                            int64_t K, int64_t total_rows, void* stream);
int vh_group_gemm_nkp_bf16(const uint16_t* A, const uint16_t* B, uint16_t* C,
                           const int64_t* cumsum, int G, int64_t N, int64_t K,
                           int64_t total_rows, void* stream);

/* Register-staged 256-square ring variants (auto-dispatched):
 * dgrad8 = !trans_b semantics; mn8 = wgrad (A^T B) semantics. */
int vh_group_gemm_dgrad8_bf16(const uint16_t* A, const uint16_t* B,
                              uint16_t* C, const int64_t* cumsum, int G,
                              int64_t N, int64_t K, int64_t total_rows,
                              void* stream);
int vh_group_gemm_mn8_bf16(const uint16_t* A, const uint16_t* B, uint16_t* C,
                           const int64_t* cumsum, int G, int64_t M, int64_t N,
                           void* stream);

/* Per-group wgrad: C[g] = A_g^T @ B_g with per-group row count (k) from
 * cumsum; A: [rows, M], B: [rows, N], C: [G, M, N] bf16 (fp32 accum).
 * Groups with zero rows are zero-filled.
 * Replaces: kernel/group_gemm.py:252-397 (group_gemm_same_mn_kernel,
 * transpose_a=True, transpose_b=False — the only variant on the path). */
int vh_group_gemm_mn_bf16(const uint16_t* A, const uint16_t* B, uint16_t* C,
                          const int64_t* cumsum, int G, int64_t M, int64_t N,
                          void* stream);

/* Batched expert-weight transpose [E, M, N] -> [E, N, M] bf16 (dgrad W^T;
 * M, N 64-multiples). */
int vh_wtranspose_bf16(const uint16_t* src, uint16_t* dst, int E, int64_t M,
                       int64_t N, void* stream);

/* Transpose-pad: [rows, C] (rows grouped by cumsum) -> [C, padded_total]
 * with each group's rows zero-padded to a 64-multiple (padded_cumsum,
 * host-computed ceil64 cumsum). Regions past the last group are untouched. */
int vh_transpose_pad_bf16(const uint16_t* src, uint16_t* dst,
                          const int64_t* cumsum, const int64_t* padded_cumsum,
                          int G, int64_t C, int64_t padded_total, void* stream);

/* wgrad over transpose-padded operands: C[g] = A'[:, pg] @ B'[:, pg]^T,
 * A' [M, PR], B' [N, PR]; group g owns padded-row range pg (64-multiples). */
int vh_group_gemm_wg256_bf16(const uint16_t* A, const uint16_t* B, uint16_t* C,
                             const int64_t* padded_cumsum, int G, int64_t M,
                             int64_t N, int64_t PR, void* stream);

/* ---- Fused MoE elementwise --------------------------------------------- */

/* Fused epilogue: act = silu(gate) * up * w_row, where gate/up are the two
 * halves of fc1 [rows, 2I] and w_row is the per-row routing weight.
 * Saves nothing; backward recomputes silu (ref moe_layer.py:328).
 * fc1: [rows, 2I] bf16; w: fp32-or-bf16 per-row [rows]; out: [rows, I] bf16. */
int vh_moe_silu_mul_weighted_bf16(const uint16_t* fc1, const uint16_t* w_row,
                                  uint16_t* out, int64_t rows, int64_t I,
                                  int has_w, void* stream);

/* Backward of the fused epilogue:
 * given dy [rows, I], fc1 [rows,2I] (pre-activation, saved), w_row,
 * produce dfc1 [rows, 2I] and (optional) dw_row [rows] fp32
 * (dw_row[t] = sum_i silu(g)*u * dy — the routing-weight grad numerator). */
int vh_moe_silu_mul_weighted_bwd_bf16(const uint16_t* dy, const uint16_t* fc1,
                                      const uint16_t* w_row, uint16_t* dfc1,
                                      float* dw_row, int64_t rows, int64_t I,
                                      int has_w, void* stream);

/* ---- RMSNorm ------------------------------------------------------------ */

/* y = w * cast_bf16(x_f32 * rsqrt(mean(x^2)+eps)); saves rstd fp32 [T].
 * Replaces the Liger RMSNorm slot (ref ops/liger/__init__.py:28-60); math
 * order matches eager Qwen3MoeRMSNorm (patched modeling :380-400). */
int vh_rmsnorm_fwd_bf16(const uint16_t* x, const uint16_t* w, uint16_t* y,
                        float* rstd, int64_t T, int64_t H, float eps,
                        void* stream);

/* Backward: dx bf16 [T,H]; dw accumulated fp32 [H] (caller zeroes dw). */
int vh_rmsnorm_bwd_bf16(const uint16_t* dy, const uint16_t* x,
                        const uint16_t* w, const float* rstd, uint16_t* dx,
                        float* dw, int64_t T, int64_t H, void* stream);

/* ---- RoPE --------------------------------------------------------------- */

/* In-place-capable rotate-half RoPE on [B, h, S, D] q and k with cos/sin
 * [B, S, D] (halves duplicated). backward = same call with negated sin
 * (host passes sign=-1). Replaces Liger RoPE slot (liger/__init__.py:115-126);
 * math matches patched modeling :94-111. */
int vh_rope_bf16(const uint16_t* q, const uint16_t* k, const uint16_t* cos_t,
                 const uint16_t* sin_t, uint16_t* q_out, uint16_t* k_out,
                 int64_t B, int64_t hq, int64_t hk, int64_t S, int64_t D,
                 int negate_sin, void* stream);

/* ---- SwiGLU ------------------------------------------------------------- */

/* out = silu(gate) * up  (bf16; fp32 internal).
 * Replaces LigerSiLUMulFunction slot (liger/__init__.py:130-141). */
int vh_silu_mul_bf16(const uint16_t* gate, const uint16_t* up, uint16_t* out,
                     int64_t n, void* stream);

/* dgate = dy * up * silu'(gate); dup = dy * silu(gate). */
int vh_silu_mul_bwd_bf16(const uint16_t* dy, const uint16_t* gate,
                         const uint16_t* up, uint16_t* dgate, uint16_t* dup,
                         int64_t n, void* stream);

/* ---- Flash attention (causal, GQA, D = 128) ----------------------------- */

/* Forward: O [B,Hq,S,128] bf16, LSE [B,Hq,S] fp32; S % 256 == 0.
 * Replaces the external flash_attn wheel behind the reference's attention
 * slot (attention/flash.py:153-301) for the packed causal path.
 * doc_start (nullable, requires B == 1): int32 [S] per-token document start
 * indices (doc_start[t] = cu_seqlens[i] for t in document i) selecting the
 * packed-varlen block-diagonal causal mask — the reference's flash-attn
 * varlen cu_seqlens path (flash.py:61-91, data_collator.py:50). */
int vh_attn_fwd_bf16(const uint16_t* Q, const uint16_t* K, const uint16_t* V,
                     uint16_t* O, float* LSE, int B, int Hq, int Hkv,
                     int64_t S, float scale, const int32_t* doc_start,
                     void* stream);

/* Backward preprocess: delta[r] = rowsum(dO[r] * O[r]), lse2[r] = LSE[r]*log2e
 * over rows = B*Hq*S flattened. */
int vh_attn_bwd_pre_bf16(const uint16_t* dO, const uint16_t* O,
                         const float* LSE, float* delta, float* lse2,
                         int64_t rows, void* stream);

/* Backward (monolithic PROBE variant): dQacc [B,Hq,S,128] fp32 (caller
 * zero-fills; atomically accumulated), dK/dV [B,Hq,S,128] bf16 per Q-head
 * (caller sums GQA groups). delta/lse2 from the preprocess. */
int vh_attn_bwd_bf16(const uint16_t* Q, const uint16_t* K, const uint16_t* V,
                     const uint16_t* dO, const float* delta, const float* lse2,
                     float* dQacc, uint16_t* dK, uint16_t* dV, int B, int Hq,
                     int Hkv, int64_t S, float scale, void* stream);

/* Split backward (the dispatched path): GQA-folded dk/dv kernel (block owns
 * 128 kv rows of one KV head and loops the whole Hq/Hkv head group — dK/dV
 * are [B,Hkv,S,128] bf16 written once, no per-Q-head intermediates) + dq
 * kernel (block owns 128 q rows, dQ [B,Hq,S,128] bf16 written once — no
 * atomics, no fp32 accumulator). doc_start/doc_end (nullable together,
 * require B == 1): int32 [S] per-token document bounds for the packed-varlen
 * block-diagonal causal mask (see vh_attn_fwd_bf16). */
int vh_attn_bwd2_bf16(const uint16_t* Q, const uint16_t* K, const uint16_t* V,
                      const uint16_t* dO, const float* delta,
                      const float* lse2, uint16_t* dQ, uint16_t* dK,
                      uint16_t* dV, int B, int Hq, int Hkv, int64_t S,
                      float scale, const int32_t* doc_start,
                      const int32_t* doc_end, void* stream);

/* ---- Fused optimizer ----------------------------------------------------- */

/* One-sweep AdamW over a device pointer table (torch semantics: decoupled
 * weight decay, bias correction; fp32 math, bf16 p/g/m/v storage). prefix is
 * the inclusive element prefix sum [T+1] (all sizes 8-multiples); grad_scale
 * (nullable fp32) divides grads in-register (grad-clip fold). */
int vh_adamw_bf16(const uint64_t* p_ptrs, const uint64_t* g_ptrs,
                  const uint64_t* m_ptrs, const uint64_t* v_ptrs,
                  const int64_t* prefix, int T, int64_t total, float lr,
                  float beta1, float beta2, float eps, float weight_decay,
                  int step, const float* grad_scale, void* stream);

/* ---- Fused chunked cross-entropy ---------------------------------------- */

/* Per-row softmax CE over a bf16 logits chunk:
 *   loss_rows[r]  = is_valid * (logsumexp(logits[r]) - logits[r][label]);
 *   dlogits[r][v] = is_valid * (softmax - onehot) * grad_scale.
 * fp32 online logsumexp; ignored rows (label == ignore_index) produce 0.
 * Replaces the inner loop of chunk_loss (ref chunk_loss.py:109-127 +
 * transformers fixed_cross_entropy). */
int vh_ce_fwd_bf16(const uint16_t* logits, const int64_t* labels,
                   float* loss_rows, uint16_t* dlogits, int64_t rows,
                   int64_t V, float grad_scale, int64_t ignore_index,
                   void* stream);

#ifdef __cplusplus
}
#endif

#endif /* VEOMNI_HIP_H */
