// C ABI between the torch binding layer (bindings.cpp, compiled by g++)
// and the gfx950 kernel library (*.hip, compiled by hipcc).
// Raw pointers + hipStream_t only -- no torch types cross this line.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

namespace pa {

// dtype tags for the few we support natively
enum DType : int { kBF16 = 0, kF32 = 1, kF16 = 2 };

// ---- norms ----------------------------------------------------------------
void layer_norm_fwd(const void* x, const void* w, const void* b, void* y,
                    float* mean, float* rstd, int64_t n, int64_t d, float eps,
                    int dtype, hipStream_t s);
void layer_norm_bwd_dx(const void* dy, const void* x, const void* w,
                       const float* mean, const float* rstd, void* dx,
                       int64_t n, int64_t d, int dtype, hipStream_t s);
int ln_dwdb_chunks(int64_t n, int64_t d);
void layer_norm_bwd_dwdb(const void* dy, const void* x, const float* mean,
                         const float* rstd, float* dw, float* db, int64_t n,
                         int64_t d, int dtype, hipStream_t s,
                         float* ws = nullptr);

void rms_norm_fwd(const void* x, const void* residual, const void* w, void* y,
                  void* res_out, float* rstd, int64_t n, int64_t d, float eps,
                  int dtype, hipStream_t s);
void rms_norm_bwd_dx(const void* dy, const void* x, const void* w,
                     const float* rstd, void* dx, int64_t n, int64_t d,
                     int dtype, hipStream_t s);
void rms_norm_bwd_dw(const void* dy, const void* x, const float* rstd,
                     float* dw, int64_t n, int64_t d, int dtype, hipStream_t s,
                     float* ws = nullptr);

// ---- fused softmax cross-entropy ------------------------------------------
// logits [n, v]; labels int64 [n]; loss/lse fp32 [n]
void softmax_ce_fwd(const void* logits, const int64_t* labels, float* loss,
                    float* lse, int64_t n, int64_t v, int64_t ignore_index,
                    int dtype, hipStream_t s);
// dlogits written in logits dtype; dloss fp32 [n]
void softmax_ce_bwd(const float* dloss, const void* logits,
                    const int64_t* labels, const float* lse, void* dlogits,
                    int64_t n, int64_t v, int64_t ignore_index, int dtype,
                    hipStream_t s);

// ---- elementwise fusions ---------------------------------------------------
// y = gelu(x + bias); bias [d] may be null
void bias_gelu_fwd(const void* x, const void* bias, void* y, int64_t n,
                   int64_t d, int dtype, hipStream_t s);
// dx = gelu'(x+bias) * dy  (dbias reduced by caller or dwdb-style kernel)
void bias_gelu_bwd(const void* dy, const void* x, const void* bias, void* dx,
                   int64_t n, int64_t d, int dtype, hipStream_t s);

// swiglu: x [n, 2d] = [gate | up]; y [n, d] = silu(gate) * up
void swiglu_fwd(const void* x, void* y, int64_t n, int64_t d, int dtype,
                hipStream_t s);
void swiglu_bwd(const void* dy, const void* x, void* dx, int64_t n, int64_t d,
                int dtype, hipStream_t s);

// rope (neox rotate-half): qk [b, s, h, dh], cos/sin fp32 [s, dh/2]
// backward == forward with conj=true (sin negated)
void rope_fwd(const void* x, const float* cos_t, const float* sin_t, void* y,
              int64_t b, int64_t sl, int64_t h, int64_t dh, int64_t pos_offset,
              bool conj, int dtype, hipStream_t s);

// column-sum of [n, d] into fp32 [d] (bias grads)
void amax_abs(const void* x, float* out, int64_t n, int dtype, hipStream_t s);
void colsum(const void* x, float* out, int64_t n, int64_t d, int dtype,
            hipStream_t s);

// ---- fused AdamW (flat shards; fp32 master + bf16 model params) -----------
void adamw(float* master, void* param_bf16, const void* grad, float* m,
           float* v, int64_t numel, float lr, float beta1, float beta2,
           float eps, float wd, float beta1_pow, float beta2_pow,
           float grad_scale, int grad_dtype, bool param_out_bf16, hipStream_t s);

// multi-tensor l2-norm^2 of a flat fp32/bf16 buffer -> out[0] (fp32, add)
void l2norm_sq(const void* x, float* out, int64_t numel, int dtype,
               hipStream_t s);

// ---- flash attention (bf16, head_dim 128 or 64) ----------------------------
// Logical layout [b, h, s, dh]; arbitrary (batch, head, seq) ELEMENT strides
// per tensor (d must be innermost/contiguous, 16B-aligned rows) so packed
// [b, s, 3, h, dh] qkv views need no transpose copies.  Strides are int64[3]
// = {batch, head, seq}.  lse/delta fp32 [b, h, sq] contiguous.
void flash_attn_fwd(const void* q, const void* k, const void* v, void* o,
                    float* lse, int64_t b, int64_t h, int64_t hkv, int64_t sq,
                    int64_t skv, int64_t dh, float scale, bool causal,
                    const int64_t* qs, const int64_t* ks, const int64_t* os,
                    hipStream_t s);
// 32x32-MFMA forward variant (same contract)
// mask: additive bf16 [B, 1|H, Sq, Skv] (ms = {m_sb, m_sh, m_sq} strides,
// null = none); pdrop + (seed, offset): Philox attention dropout
// (dropout_impl.cu.h:129 seed/offset semantics, recompute-deterministic)
void flash_attn_fwd32(const void* q, const void* k, const void* v, void* o,
                      float* lse, int64_t b, int64_t h, int64_t hkv, int64_t sq,
                      int64_t skv, int64_t dh, float scale, bool causal,
                      const int64_t* qs, const int64_t* ks, const int64_t* os,
                      const void* mask, const int64_t* ms, float pdrop,
                      uint64_t seed, uint64_t offset, hipStream_t s);
void flash_attn_bwd_dq32(const void* dout, const void* q, const void* k,
                         const void* v, const float* lse, const float* delta,
                         void* dq, int64_t b, int64_t h, int64_t sq, int64_t skv,
                         int64_t dh, float scale, bool causal,
                         const int64_t* qs, const int64_t* ks, const int64_t* dos,
                         const int64_t* dqs, const void* mask, const int64_t* ms,
                         float pdrop, uint64_t seed, uint64_t offset,
                         hipStream_t s);
void flash_attn_bwd_dkv32(const void* dout, const void* q, const void* k,
                          const void* v, const float* lse, const float* delta,
                          void* dk, void* dv, int64_t b, int64_t h, int64_t sq,
                          int64_t skv, int64_t dh, float scale, bool causal,
                          const int64_t* qs, const int64_t* ks, const int64_t* dos,
                          const int64_t* dks, const void* mask, const int64_t* ms,
                          float pdrop, uint64_t seed, uint64_t offset,
                          hipStream_t s);
void flash_attn_bwd(const void* dout, const void* q, const void* k,
                    const void* v, const void* o, const float* lse,
                    void* dq, void* dk, void* dv, float* delta, int64_t b,
                    int64_t h, int64_t hkv, int64_t sq, int64_t skv,
                    int64_t dh, float scale, bool causal,
                    const int64_t* qs, const int64_t* ks, const int64_t* dos,
                    const int64_t* os, const int64_t* dqs, const int64_t* dks,
                    const void* mask, const int64_t* ms, float pdrop,
                    uint64_t seed, uint64_t offset, hipStream_t s);

// ---- dropout + residual add ------------------------------------------------
// y = dropout(x, p) + residual; mask stored as uint8 per element
void dropout_add_fwd(const void* x, const void* residual, void* y,
                     uint8_t* mask, int64_t numel, float p, uint64_t seed,
                     uint64_t offset, int dtype, hipStream_t s);
void dropout_add_bwd(const void* dy, const uint8_t* mask, void* dx,
                     int64_t numel, float p, int dtype, hipStream_t s);

// ---- embedding -------------------------------------------------------------
void embedding_fwd(const void* table, const int64_t* ids, void* out,
                   int64_t n_ids, int64_t d, int64_t vocab, int64_t padding_idx,
                   int dtype, hipStream_t s);
// grad table accumulated fp32 (atomics)
void embedding_bwd(const void* dout, const int64_t* ids, float* dtable,
                   int64_t n_ids, int64_t d, int64_t vocab, int64_t padding_idx,
                   int dtype, hipStream_t s);

// ---- paged-KV decode attention (serving) ----------------------------------
// blk/pos/head strides in elements (0,0,0 = default paged layout
// [nblocks, bs, HKV, D]); dense [B, H, S, D] caches pass their own
void decode_attention(const void* q, const void* kcache, const void* vcache,
                      const int* block_table, const int* seq_lens, void* o,
                      int64_t b, int64_t h, int64_t hkv, int64_t bs,
                      int64_t max_blocks, int64_t dh, float scale,
                      int64_t blk_str, int64_t pos_str, int64_t head_str,
                      hipStream_t s);

// ---- hand-written bf16 MFMA GEMM (C[m][n] = op(A) x op(B)) ----------------
// b_is_nt: B passed as Bt[n][k] row-major (fast path); else B[k][n].
void gemm_bf16_8p(const void* a, const void* b, void* c, int64_t m, int64_t n,
                  int64_t k, int64_t lda, int64_t ldb, int64_t ldc,
                  hipStream_t s);
void gemm_bf16_nt_batched(const void* a, const void* b, void* c,
                          int64_t batch, int64_t m, int64_t n, int64_t k,
                          int64_t lda, int64_t ldb, int64_t ldc,
                          int64_t a_bs, int64_t b_bs, int64_t c_bs,
                          hipStream_t s);
void gemm_bf16(const void* a, const void* b, void* c, int64_t m, int64_t n,
               int64_t k, int64_t lda, int64_t ldb, int64_t ldc, bool b_is_nt,
               hipStream_t s);

// full variant with operand layout + fused epilogue:
//   layout: 0=NT (A[m][k], Bt[n][k])  1=NN (A[m][k], B[k][n])
//           2=TN (At[k][m], B[k][n])  -- wgrad
//   epilogue: 0=none 1=+bias[n] 2=gelu(x+bias) writing pre-act to aux
//             3=dgelu (out = C * gelu'(aux))
//   accumulate: C += result (epilogue forced to none)
void gemm_bf16_ex(const void* a, const void* b, void* c, const void* bias,
                  void* aux, int64_t m, int64_t n, int64_t k, int64_t lda,
                  int64_t ldb, int64_t ldc, int layout, int epilogue,
                  bool accumulate, hipStream_t s);

// weight-only int8 GEMV (decode): qweight [N,K] int8 rows, scale [N] fp32
void weight_only_gemv(const void* x, const void* wq, const float* scale,
                      const void* bias, void* out, int64_t m, int64_t n,
                      int64_t k, int dtype, hipStream_t s);

// varlen (flash_attn_unpadded): packed [total, H, D], ONE launch; bmap =
// int32 (seq, offset-within-seq) pairs per block; lse/delta [H, total_q]
void flash_attn_varlen_fwd32(const void* q, const void* k, const void* v,
                             void* o, float* lse, int64_t h, int64_t hkv,
                             int64_t total_q, int64_t total_k, int64_t dh,
                             float scale, bool causal, int64_t nblocks,
                             const int* cu_q, const int* cu_k, const int* bmap,
                             float pdrop, uint64_t seed, uint64_t offset,
                             hipStream_t s);
void flash_attn_varlen_bwd32(const void* dout, const void* q, const void* k,
                             const void* v, const float* lse, const float* delta,
                             void* dq, void* dk, void* dv, int64_t h,
                             int64_t total_q, int64_t total_k, int64_t dh,
                             float scale, bool causal, int64_t nqblocks,
                             int64_t nkvblocks, const int* cu_q, const int* cu_k,
                             const int* qbmap, const int* kvbmap, float pdrop,
                             uint64_t seed, uint64_t offset, hipStream_t s);
void flash_attn_varlen_bwd(const void* dout, const void* q, const void* k,
                           const void* v, const void* o, const float* lse,
                           void* dq, void* dk, void* dv, float* delta,
                           int64_t h, int64_t total_q, int64_t total_k,
                           int64_t dh, float scale, bool causal,
                           int64_t nqblocks, int64_t nkvblocks,
                           const int* cu_q, const int* cu_k, const int* qbmap,
                           const int* kvbmap, float pdrop, uint64_t seed,
                           uint64_t offset, hipStream_t s);

// materialize the dropout keep-mask (debug/tests): out uint8 [total]
void fa_dropout_mask(void* out, int64_t total, float p, uint64_t seed,
                     uint64_t offset, hipStream_t s);

// fp8 e4m3 MX GEMM: C[m,n] (bf16) = scale_ab * (A_fp8[m,k] x Bt_fp8[n,k]^T)
void gemm_fp8_nt(const void* a, const void* bt, void* c, const void* bias,
                 float scale_ab, int64_t m, int64_t n, int64_t k, int64_t lda,
                 int64_t ldb, int64_t ldc, hipStream_t s);
// grouped/batched variant (one launch over all experts; *_bs = batch strides)
void gemm_fp8_nt_batched(const void* a, const void* bt, void* c,
                         const void* bias, float scale_ab, int64_t batch,
                         int64_t m, int64_t n, int64_t k, int64_t lda,
                         int64_t ldb, int64_t ldc, int64_t a_bs, int64_t b_bs,
                         int64_t c_bs, hipStream_t s,
                         int64_t bias_bs = 0);
// bf16 -> e4m3 cast with uniform scale (fused, no fp32 round-trip)
void quant_fp8(const void* x, void* out, float scale, int64_t numel,
               hipStream_t s);

// skinny decode GEMM (M<=32): y = x @ W[K,N] + bias, split-K partials
// in `workspace` ([ksplit, padded_m, N] fp32)
void decode_gemm(const void* x, const void* w, const void* bias, void* y,
                 float* workspace, int64_t m, int64_t n, int64_t k,
                 int64_t ldw, int64_t ksplit, hipStream_t s);
void decode_gemm_mfma(const void* x, const void* w, const void* bias, void* y,
                 float* workspace, int64_t m, int64_t n, int64_t k,
                 int64_t ldw, int64_t ksplit, hipStream_t s,
                 const float* chscale = nullptr, bool int8w = false);

// ---- MoE routing (assign_pos/number_count/gate parity) --------------------
void moe_gate_topk(const float* logits, float* topv, int* topi, float* me,
                   float* ce, int64_t t, int64_t e, int64_t k, hipStream_t s);
void moe_assign_slots(const int* topi, int* slot_of, int* counts, int64_t t,
                      int64_t e, int64_t k, int64_t cap, hipStream_t s);

}  // namespace pa
