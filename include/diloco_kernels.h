/* diloco_kernels.h — C-ABI boundary of the MI355X-native DiLoCo hot path.
 *
 * Every entry point here replaces a GPU-math call site of the reference
 * (PrimeIntellect-ai/OpenDiloco @ 2024-10-08); the reference interface each
 * one replaces is cited per function.  Conventions:
 *   - all device pointers are raw, caller-owned, stream-ordered;
 *   - `stream` is a hipStream_t passed as void*;
 *   - return value is the hipError_t of the launch (0 == success);
 *   - no internal synchronisation, no allocation;
 *   - dk_dtype: 0 = float32, 1 = float16, 2 = bfloat16.
 *
 * The Python side (opendiloco_amd/ops.py) binds these through a thin torch
 * extension (csrc/binding.cpp); a non-torch host could bind them with ctypes
 * or dlopen directly.
 */
#ifndef DILOCO_KERNELS_H
#define DILOCO_KERNELS_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef void* dkStream; /* hipStream_t */

enum dk_dtype { DK_F32 = 0, DK_F16 = 1, DK_BF16 = 2 };

/* ---- RMSNorm ------------------------------------------------------------
 * Replaces transformers LlamaRMSNorm fwd/bwd, invoked from the model
 * forward the reference calls at train_fsdp.py:378 / train_diloco_torch.py:313.
 * y[r,c] = w[c] * (x[r,c] * invrms[r]);  invrms[r] = rsqrt(mean(x[r,:]^2)+eps)
 * x,y,w: dtype; invrms: f32[rows] (saved for bwd). */
/* res/h_out (nullable, paired): fuse the preceding residual add
 * h = x + res (the reference's `x + self_attn(...)` adds inside the decoder
 * layer) into the normalisation pass: h is written out, y = norm(h). */
int dk_rmsnorm_fwd(void* y, void* h_out, float* invrms, const void* x,
                   const void* res, const void* w, int64_t rows, int64_t cols,
                   float eps, int dtype, dkStream stream);
/* dx: dtype; dres (nullable): upstream residual gradient added into dx
 * (dx == d(residual input) == d(res)); dw_partial: f32[grid][cols] workspace
 * written deterministically; grid from dk_rmsnorm_bwd_grid(rows). A second
 * call to dk_reduce_partials sums dw_partial into dw (f32[cols]). */
int dk_rmsnorm_bwd(void* dx, float* dw_partial, const void* dy, const void* dres,
                   const void* x, const void* w, const float* invrms,
                   int64_t rows, int64_t cols, int grid, int dtype, dkStream stream);
int dk_rmsnorm_bwd_grid(int64_t rows);
/* out[c] = sum_g partial[g][c], fixed order (deterministic), two parallel
 * stages through tmp (f32[dk_reduce_tmp_slices(grid)][cols]). */
int dk_reduce_tmp_slices(int grid);
int dk_reduce_partials(float* out, float* tmp, const float* partial, int grid,
                       int64_t cols, dkStream stream);

/* ---- RoPE ---------------------------------------------------------------
 * Replaces transformers' apply_rotary_pos_emb (half-split convention:
 * rotate_half(x) = cat(-x2, x1)) inside the reference's model forward.
 * x: [n_tok_rows, D] where a row is one (b, h, s) head-vector; costab/sintab:
 * f32[S, D/2]; row r's position = (r / heads_stride) % S ... simplified:
 * caller passes seq len S and heads H so pos = (r % (S*?)).  We use layout
 * [B, H, S, D]: pos = (r % S).  backward=1 applies the transposed rotation. */
int dk_rope(void* out, const void* x, const float* costab, const float* sintab,
            int64_t n_rows, int64_t S, int64_t D, int backward, int dtype,
            dkStream stream);
/* generic-stride RoPE gather/scatter: moves a head-tensor between the fused
 * QKV buffer layout (addr = b*sb + h*sh + s*sr) and contiguous [B,H,S,D],
 * applying the rotation (rotate=1; backward=1 -> transposed) or plain copy
 * (rotate=0, the V path) in the same pass. */
int dk_rope_move(void* out, const void* in, const float* costab, const float* sintab,
                 int64_t B, int64_t H, int64_t S, int64_t D,
                 int64_t i_sb, int64_t i_sh, int64_t i_sr,
                 int64_t o_sb, int64_t o_sh, int64_t o_sr,
                 int backward, int rotate, int dtype, dkStream stream);

/* ---- SwiGLU -------------------------------------------------------------
 * Replaces transformers LlamaMLP's act_fn(gate)*up (silu).
 * y = silu(gate) * up, elementwise over n elements. */
int dk_swiglu_fwd(void* y, const void* gate, const void* up, int64_t n,
                  int dtype, dkStream stream);
int dk_swiglu_bwd(void* dgate, void* dup, const void* dy, const void* gate,
                  const void* up, int64_t n, int dtype, dkStream stream);
/* fused-layout variant over the batched gate-up GEMM output gu [rows, 2*I]:
 * y[r,c] = silu(gu[r,c]) * gu[r,I+c]; bwd writes dgu in the same layout. */
int dk_swiglu2_fwd(void* y, const void* gu, int64_t rows, int64_t I, int dtype,
                   dkStream stream);
int dk_swiglu2_bwd(void* dgu, const void* dy, const void* gu, int64_t rows,
                   int64_t I, int dtype, dkStream stream);

/* ---- Fused cross-entropy ------------------------------------------------
 * Replaces the causal-LM loss of LlamaForCausalLM (logits->fp32, shifted CE,
 * mean over tokens) invoked via model(**batch) at train_fsdp.py:378.  The
 * causal shift is internal: logits dtype[B, S, V], labels int64[B, S];
 * position s < S-1 scores against labels[b, s+1]; loss_rows/lse are
 * f32[B*(S-1)]. */
int dk_cross_entropy_fwd(float* loss_rows, float* lse, const void* logits,
                         const int64_t* labels, int64_t B, int64_t S, int64_t V,
                         int dtype, dkStream stream);
/* dlogits (full [B, S, V], last position zeroed) =
 * (softmax(logits) - onehot(shifted labels)) * (*dloss) * inv_T. */
int dk_cross_entropy_bwd(void* dlogits, const void* logits, const float* lse,
                         const int64_t* labels, const float* dloss, float inv_T,
                         int64_t B, int64_t S, int64_t V, int dtype, dkStream stream);

/* ---- Flash attention (causal, GQA) --------------------------------------
 * Replaces torch SDPA inside the reference's model forward (attn_implementation
 * "sdpa", train_fsdp.py:107).  Layout [B, H, S, D] contiguous; D in {32,64};
 * o: dtype; lse: f32[B,Hq,S] = m + log(l) saved for bwd.  scale = 1/sqrt(D). */
/* o_sb/o_sh/o_sr (and g_* on the backward entries): stride triplet
 * (addr = b*sb + h*sh + s*sr) for the attention output / upstream gradient,
 * so o can be written directly in [B, S, Hq*D] and do read from it without
 * transpose copies; sb == 0 selects the contiguous [B,H,S,D] default.
 * v_sb/v_sh/v_sr (and dv_* on dk_attn_bwd_dkdv): the same triplet for V /
 * its gradient, so V can be read straight out of the packed QKV projection
 * [B, S, (Hq+2*Hkv)*D] and dV written back into the packed gradient without
 * gather/scatter passes (V carries no RoPE rotation).  Rows must stay
 * 16-byte aligned (strides in elements, last dim contiguous).  dv_* only
 * makes sense when Hq == Hkv (no GQA group summation). */
int dk_attn_fwd(void* o, float* lse, const void* q, const void* k, const void* v,
                int64_t B, int64_t Hq, int64_t Hkv, int64_t S, int64_t D,
                float scale, int64_t o_sb, int64_t o_sh, int64_t o_sr,
                int64_t v_sb, int64_t v_sh, int64_t v_sr,
                int dtype, dkStream stream);
/* delta[b,h,s] = rowsum(do * o), fp32 — preprocessing for bwd. */
int dk_attn_bwd_preprocess(float* delta, const void* do_, const void* o,
                           int64_t B, int64_t H, int64_t S, int64_t D,
                           int64_t g_sb, int64_t g_sh, int64_t g_sr,
                           int dtype, dkStream stream);
int dk_attn_bwd_dkdv(void* dk_out, void* dv_out, const void* do_, const void* q,
                     const void* k, const void* v, const float* lse,
                     const float* delta, int64_t B, int64_t Hq, int64_t Hkv,
                     int64_t S, int64_t D, float scale,
                     int64_t g_sb, int64_t g_sh, int64_t g_sr,
                     int64_t v_sb, int64_t v_sh, int64_t v_sr,
                     int64_t dv_sb, int64_t dv_sh, int64_t dv_sr,
                     int dtype, dkStream stream);
int dk_attn_bwd_dq(void* dq_out, const void* do_, const void* q, const void* k,
                   const void* v, const float* lse, const float* delta,
                   int64_t B, int64_t Hq, int64_t Hkv, int64_t S, int64_t D,
                   float scale, int64_t g_sb, int64_t g_sh, int64_t g_sr,
                   int64_t v_sb, int64_t v_sh, int64_t v_sr,
                   int dtype, dkStream stream);

/* ---- Fused AdamW (inner optimizer) --------------------------------------
 * Replaces the torch AdamW step the reference runs at hivemind_diloco.py:546-550
 * and train_diloco_torch.py:186,325 (lr cfg, wd=0.1, betas=(0.9,0.95)).
 * Flat fp32 buffers p, g, m, v of n elements; torch single-tensor op order
 * (mul_(1-lr*wd); lerp_(m,g,1-b1); v=b2*v+(1-b2)g^2; p -= lr/bc1 * m/(sqrt(v)/sqrt(bc2)+eps)). */
int dk_fused_adamw(float* p, const float* g, float* m, float* v,
                   int64_t n, float lr, float beta1, float beta2, float eps,
                   float weight_decay, int step, dkStream stream);

/* ---- Gradient clipping --------------------------------------------------
 * Replaces clip_grad_norm_(1.0) at train_fsdp.py:395 / train_diloco_torch.py:323.
 * Two deterministic passes: partial sums of squares, then finalize+scale.
 * partials: f32[grid]; grid from dk_gradsq_grid(n). */
int dk_grad_sq_partials(float* partials, const float* g, int64_t n, int grid,
                        dkStream stream);
int dk_gradsq_grid(int64_t n);
/* total_norm_out (f32 device scalar) = sqrt(sum partials); then
 * g *= min(1, max_norm / (total_norm + 1e-6))   (torch semantics). */
int dk_clip_apply(float* g, float* total_norm_out, const float* partials,
                  int grid, int64_t n, float max_norm, dkStream stream);

/* ---- Outer step (pseudo-gradient + Nesterov SGD + copy-back) ------------
 * Replaces: pseudo-grad theta_outer - theta_local (hivemind_diloco.py:158-167,
 * train_diloco_torch.py:342-344). Flat fp32. */
int dk_pseudo_grad(float* g_out, const float* theta_outer, const float* theta_local,
                   int64_t n, dkStream stream);
/* After all-reduce of g: Nesterov SGD on theta_outer (torch SGD math:
 * buf = mu*buf + g (or buf=g when first=1); d = g + mu*buf;
 * theta_outer -= lr*d) then theta_local = theta_outer (copy-back,
 * hivemind_diloco.py:654-665,716-720; train_diloco_torch.py:346-353). */
int dk_outer_nesterov(float* theta_outer, float* theta_local, float* momentum_buf,
                      const float* g_avg, int64_t n, float lr, float momentum,
                      int first_step, dkStream stream);

/* ---- misc ---------------------------------------------------------------
 * Elementwise scaled cast between f32 and bf16/f16 flat buffers (used for
 * fp16/bf16 all-reduce payloads when compression is requested). */
/* fp32 dst[i] += toF(src[i]) — fused low-precision dW accumulation into the
 * fp32 master gradient (replaces the reference's autocast cast+add pair,
 * train_diloco_torch.py:305-310 grad flow). */
int dk_accum(float* dst, const void* src, int64_t n, int src_dtype,
             dkStream stream);

/* fp32 dst[i] += sum_b src[b*chunk_stride + i] — deterministic reduction of
 * the split-K weight-gradient partial slabs into the fp32 master gradient
 * (the dW = dy^T x GEMMs of every projection, reference F.linear backward
 * inside transformers Llama at train_fsdp.py:383; split over the token
 * dimension to fill all 256 CUs). */
int dk_accum_chunks(float* dst, const float* src, int64_t n, int nchunk,
                    int64_t chunk_stride, dkStream stream);
int dk_cast(void* dst, const void* src, int64_t n, int dst_dtype, int src_dtype,
            dkStream stream);

/* MFMA fragment-layout probes (test-only): writes, for each lane l and slot j,
 * the (row, k) element index that slot maps to, by multiplying basis matrices.
 * kind: 0 = A-layout of mfma_f32_16x16x32_bf16, 1 = B-layout, 2 = C-layout. */
int dk_probe_mfma_16x16x32_bf16(float* out_d, const void* a16x32, const void* b32x16,
                                dkStream stream);
int dk_probe_mfma_16x16x32_bf16_alt(float* out_d, const void* a16x32, const void* b32x16,
                                    dkStream stream);
int dk_probe_mfma_32x32x16_bf16(float* out_d, const void* a32x16, const void* b16x32,
                                dkStream stream);
int dk_probe_permlane32(int* out_d, dkStream stream);

/* version / build info */
const char* dk_version(void);

#ifdef __cplusplus
}
#endif
#endif /* DILOCO_KERNELS_H */
