// binding.cpp — thin torch extension over the C-ABI kernel library.
//
// This file is host-only glue: tensor validation, output allocation (via the
// torch caching allocator), dtype mapping and stream plumbing.  All device
// math lives behind include/diloco_kernels.h (libdiloco_kernels.so).
// Written directly against torch's native HIP/ROCm surface (c10::hip) —
// no CUDA-compat layer.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "../../include/diloco_kernels.h"

#include <rocblas/rocblas.h>
#include <hipblaslt/hipblaslt.h>

#include <map>
#include <tuple>
#include <vector>

namespace {

int dt_of(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return DK_F32;
    case at::kHalf: return DK_F16;
    case at::kBFloat16: return DK_BF16;
    default: TORCH_CHECK(false, "unsupported dtype ", t.scalar_type());
  }
}

void* stream() {
  return (void*)c10::hip::getCurrentHIPStream().stream();
}

#define DK_OK(call)                                                        \
  do {                                                                     \
    int rc_ = (call);                                                      \
    TORCH_CHECK(rc_ == 0, #call, " failed with hip error code ", rc_);     \
  } while (0)

#define CHECK_DEV_CONTIG(t) \
  TORCH_CHECK((t).is_cuda() && (t).is_contiguous(), #t " must be a contiguous device tensor")

// ---- RMSNorm ----
std::vector<at::Tensor> rmsnorm_fwd(const at::Tensor& x, const at::Tensor& w, double eps) {
  CHECK_DEV_CONTIG(x);
  CHECK_DEV_CONTIG(w);
  const int64_t cols = x.size(-1);
  const int64_t rows = x.numel() / cols;
  auto y = at::empty_like(x);
  auto invrms = at::empty({rows}, x.options().dtype(at::kFloat));
  DK_OK(dk_rmsnorm_fwd(y.data_ptr(), nullptr, invrms.data_ptr<float>(), x.data_ptr(),
                       nullptr, w.data_ptr(), rows, cols, (float)eps, dt_of(x), stream()));
  return {y, invrms};
}

std::vector<at::Tensor> rmsnorm_add_fwd(const at::Tensor& x, const at::Tensor& res,
                                        const at::Tensor& w, double eps) {
  CHECK_DEV_CONTIG(x);
  CHECK_DEV_CONTIG(res);
  const int64_t cols = x.size(-1);
  const int64_t rows = x.numel() / cols;
  auto y = at::empty_like(x);
  auto h = at::empty_like(x);
  auto invrms = at::empty({rows}, x.options().dtype(at::kFloat));
  DK_OK(dk_rmsnorm_fwd(y.data_ptr(), h.data_ptr(), invrms.data_ptr<float>(), x.data_ptr(),
                       res.data_ptr(), w.data_ptr(), rows, cols, (float)eps, dt_of(x),
                       stream()));
  return {y, h, invrms};
}

std::vector<at::Tensor> rmsnorm_bwd(const at::Tensor& dy,
                                    const c10::optional<at::Tensor>& dres,
                                    const at::Tensor& x, const at::Tensor& w,
                                    const at::Tensor& invrms) {
  CHECK_DEV_CONTIG(dy);
  CHECK_DEV_CONTIG(x);
  const int64_t cols = x.size(-1);
  const int64_t rows = x.numel() / cols;
  const int grid = dk_rmsnorm_bwd_grid(rows);
  auto dx = at::empty_like(x);
  auto dwp = at::empty({grid, cols}, x.options().dtype(at::kFloat));
  auto tmp = at::empty({dk_reduce_tmp_slices(grid), cols}, x.options().dtype(at::kFloat));
  auto dw = at::empty({cols}, x.options().dtype(at::kFloat));
  const void* dres_p = dres.has_value() ? dres->data_ptr() : nullptr;
  DK_OK(dk_rmsnorm_bwd(dx.data_ptr(), dwp.data_ptr<float>(), dy.data_ptr(), dres_p,
                       x.data_ptr(), w.data_ptr(), invrms.data_ptr<float>(), rows, cols,
                       grid, dt_of(x), stream()));
  DK_OK(dk_reduce_partials(dw.data_ptr<float>(), tmp.data_ptr<float>(),
                           dwp.data_ptr<float>(), grid, cols, stream()));
  return {dx, dw};
}

// ---- RoPE ----
at::Tensor rope(const at::Tensor& x, const at::Tensor& costab, const at::Tensor& sintab,
                int64_t S, bool backward) {
  CHECK_DEV_CONTIG(x);
  const int64_t D = x.size(-1);
  const int64_t n_rows = x.numel() / D;
  auto out = at::empty_like(x);
  DK_OK(dk_rope(out.data_ptr(), x.data_ptr(), costab.data_ptr<float>(),
                sintab.data_ptr<float>(), n_rows, S, D, backward ? 1 : 0, dt_of(x), stream()));
  return out;
}

// ---- SwiGLU ----
at::Tensor swiglu_fwd(const at::Tensor& gate, const at::Tensor& up) {
  CHECK_DEV_CONTIG(gate);
  CHECK_DEV_CONTIG(up);
  auto y = at::empty_like(gate);
  DK_OK(dk_swiglu_fwd(y.data_ptr(), gate.data_ptr(), up.data_ptr(), gate.numel(),
                      dt_of(gate), stream()));
  return y;
}

std::vector<at::Tensor> swiglu_bwd(const at::Tensor& dy, const at::Tensor& gate,
                                   const at::Tensor& up) {
  auto dgate = at::empty_like(gate);
  auto dup = at::empty_like(up);
  DK_OK(dk_swiglu_bwd(dgate.data_ptr(), dup.data_ptr(), dy.data_ptr(), gate.data_ptr(),
                      up.data_ptr(), gate.numel(), dt_of(gate), stream()));
  return {dgate, dup};
}

at::Tensor swiglu2_fwd(const at::Tensor& gu, int64_t I) {
  CHECK_DEV_CONTIG(gu);
  const int64_t rows = gu.numel() / (2 * I);
  auto sizes = gu.sizes().vec();
  sizes.back() = I;
  auto y = at::empty(sizes, gu.options());
  DK_OK(dk_swiglu2_fwd(y.data_ptr(), gu.data_ptr(), rows, I, dt_of(gu), stream()));
  return y;
}

at::Tensor swiglu2_bwd(const at::Tensor& dy, const at::Tensor& gu, int64_t I) {
  auto dgu = at::empty_like(gu);
  const int64_t rows = gu.numel() / (2 * I);
  DK_OK(dk_swiglu2_bwd(dgu.data_ptr(), dy.data_ptr(), gu.data_ptr(), rows, I,
                       dt_of(gu), stream()));
  return dgu;
}

// ---- cross entropy ----
std::vector<at::Tensor> ce_fwd(const at::Tensor& logits, const at::Tensor& labels) {
  CHECK_DEV_CONTIG(logits);
  TORCH_CHECK(logits.dim() == 3, "ce_fwd expects [B, S, V] logits");
  const int64_t B = logits.size(0), S = logits.size(1), V = logits.size(2);
  const int64_t T = B * (S - 1);
  auto loss_rows = at::empty({T}, logits.options().dtype(at::kFloat));
  auto lse = at::empty({T}, logits.options().dtype(at::kFloat));
  DK_OK(dk_cross_entropy_fwd(loss_rows.data_ptr<float>(), lse.data_ptr<float>(),
                             logits.data_ptr(), labels.data_ptr<int64_t>(), B, S, V,
                             dt_of(logits), stream()));
  return {loss_rows, lse};
}

at::Tensor ce_bwd(const at::Tensor& logits, const at::Tensor& lse, const at::Tensor& labels,
                  const at::Tensor& dloss, double inv_T) {
  auto dlogits = at::empty_like(logits);
  const int64_t B = logits.size(0), S = logits.size(1), V = logits.size(2);
  DK_OK(dk_cross_entropy_bwd(dlogits.data_ptr(), logits.data_ptr(), lse.data_ptr<float>(),
                             labels.data_ptr<int64_t>(), dloss.data_ptr<float>(),
                             (float)inv_T, B, S, V, dt_of(logits), stream()));
  return dlogits;
}

// ---- attention ----
std::vector<at::Tensor> attn_fwd(const at::Tensor& q, const at::Tensor& k,
                                 const at::Tensor& v, double scale) {
  CHECK_DEV_CONTIG(q);
  CHECK_DEV_CONTIG(k);
  CHECK_DEV_CONTIG(v);
  const int64_t B = q.size(0), Hq = q.size(1), S = q.size(2), D = q.size(3);
  const int64_t Hkv = k.size(1);
  auto o = at::empty_like(q);
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  DK_OK(dk_attn_fwd(o.data_ptr(), lse.data_ptr<float>(), q.data_ptr(), k.data_ptr(),
                    v.data_ptr(), B, Hq, Hkv, S, D, (float)scale, 0, 0, 0, 0, 0, 0,
                    dt_of(q), stream()));
  return {o, lse};
}

std::vector<at::Tensor> attn_bwd(const at::Tensor& do_, const at::Tensor& q,
                                 const at::Tensor& k, const at::Tensor& v,
                                 const at::Tensor& o, const at::Tensor& lse, double scale) {
  CHECK_DEV_CONTIG(do_);
  const int64_t B = q.size(0), Hq = q.size(1), S = q.size(2), D = q.size(3);
  const int64_t Hkv = k.size(1);
  auto delta = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  DK_OK(dk_attn_bwd_preprocess(delta.data_ptr<float>(), do_.data_ptr(), o.data_ptr(),
                               B, Hq, S, D, 0, 0, 0, dt_of(q), stream()));
  auto dq = at::empty_like(q);
  // dk/dv computed per q-head; python caller sums GQA groups when Hq != Hkv
  auto dk_full = at::empty({B, Hq, S, D}, q.options());
  auto dv_full = at::empty({B, Hq, S, D}, q.options());
  DK_OK(dk_attn_bwd_dkdv(dk_full.data_ptr(), dv_full.data_ptr(), do_.data_ptr(),
                         q.data_ptr(), k.data_ptr(), v.data_ptr(), lse.data_ptr<float>(),
                         delta.data_ptr<float>(), B, Hq, Hkv, S, D, (float)scale,
                         0, 0, 0, 0, 0, 0, 0, 0, 0, dt_of(q), stream()));
  DK_OK(dk_attn_bwd_dq(dq.data_ptr(), do_.data_ptr(), q.data_ptr(), k.data_ptr(),
                       v.data_ptr(), lse.data_ptr<float>(), delta.data_ptr<float>(),
                       B, Hq, Hkv, S, D, (float)scale, 0, 0, 0, 0, 0, 0,
                       dt_of(q), stream()));
  return {dq, dk_full, dv_full};
}

// ---- fused QKV attention path: o in [B, S, Hq*D]; do read back the same way;
// rope gather/scatter between the packed qkv buffer and BHSD head tensors ----
at::Tensor qkv_rope_gather(const at::Tensor& qkv, const at::Tensor& costab,
                           const at::Tensor& sintab, int64_t H, int64_t D,
                           int64_t col_off, bool rotate) {
  CHECK_DEV_CONTIG(qkv);
  const int64_t B = qkv.size(0), S = qkv.size(1), TOT = qkv.size(2);
  auto out = at::empty({B, H, S, D}, qkv.options());
  DK_OK(dk_rope_move(out.data_ptr(), (const char*)qkv.data_ptr() + col_off * qkv.element_size(),
                     costab.data_ptr<float>(), sintab.data_ptr<float>(), B, H, S, D,
                     /*in*/ S * TOT, D, TOT, /*out*/ H * S * D, S * D, D,
                     0, rotate ? 1 : 0, dt_of(qkv), stream()));
  return out;
}

void rope_scatter_(at::Tensor& dqkv, const at::Tensor& src_bhsd, const at::Tensor& costab,
                   const at::Tensor& sintab, int64_t col_off, bool rotate) {
  CHECK_DEV_CONTIG(dqkv);
  CHECK_DEV_CONTIG(src_bhsd);
  const int64_t B = dqkv.size(0), S = dqkv.size(1), TOT = dqkv.size(2);
  const int64_t H = src_bhsd.size(1), D = src_bhsd.size(3);
  DK_OK(dk_rope_move((char*)dqkv.data_ptr() + col_off * dqkv.element_size(),
                     src_bhsd.data_ptr(),
                     costab.data_ptr<float>(), sintab.data_ptr<float>(), B, H, S, D,
                     /*in*/ H * S * D, S * D, D, /*out*/ S * TOT, D, TOT,
                     1, rotate ? 1 : 0, dt_of(dqkv), stream()));
}

std::vector<at::Tensor> attn_fwd_bsd(const at::Tensor& q, const at::Tensor& k,
                                     const at::Tensor& v, double scale) {
  CHECK_DEV_CONTIG(q);
  const int64_t B = q.size(0), Hq = q.size(1), S = q.size(2), D = q.size(3);
  const int64_t Hkv = k.size(1);
  // v may be a strided view into the packed QKV projection (last dim contiguous)
  TORCH_CHECK(v.stride(3) == 1, "attn_fwd_bsd: v last dim must be contiguous");
  const bool vc = v.is_contiguous();
  auto o = at::empty({B, S, Hq * D}, q.options());
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  DK_OK(dk_attn_fwd(o.data_ptr(), lse.data_ptr<float>(), q.data_ptr(), k.data_ptr(),
                    v.data_ptr(), B, Hq, Hkv, S, D, (float)scale,
                    /*o strides (b,h,s)*/ S * Hq * D, D, Hq * D,
                    vc ? 0 : v.stride(0), vc ? 0 : v.stride(1), vc ? 0 : v.stride(2),
                    dt_of(q), stream()));
  return {o, lse};
}

std::vector<at::Tensor> attn_bwd_bsd(const at::Tensor& do_bsd, const at::Tensor& q,
                                     const at::Tensor& k, const at::Tensor& v,
                                     const at::Tensor& o_bsd, const at::Tensor& lse,
                                     double scale,
                                     c10::optional<at::Tensor> dv_out = c10::nullopt) {
  CHECK_DEV_CONTIG(do_bsd);
  const int64_t B = q.size(0), Hq = q.size(1), S = q.size(2), D = q.size(3);
  const int64_t Hkv = k.size(1);
  const int64_t sb = S * Hq * D, sh = D, sr = Hq * D;
  TORCH_CHECK(v.stride(3) == 1, "attn_bwd_bsd: v last dim must be contiguous");
  const bool vc = v.is_contiguous();
  const int64_t v_sb = vc ? 0 : v.stride(0), v_sh = vc ? 0 : v.stride(1),
                v_sr = vc ? 0 : v.stride(2);
  auto delta = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  DK_OK(dk_attn_bwd_preprocess(delta.data_ptr<float>(), do_bsd.data_ptr(),
                               o_bsd.data_ptr(), B, Hq, S, D, sb, sh, sr,
                               dt_of(q), stream()));
  auto dq = at::empty_like(q);
  auto dk_full = at::empty({B, Hq, S, D}, q.options());
  at::Tensor dv_full;
  int64_t dv_sb = 0, dv_sh = 0, dv_sr = 0;
  void* dv_ptr;
  if (dv_out.has_value()) {
    // write dV straight into a strided view (e.g. the packed dQKV buffer);
    // only valid without GQA group summation
    TORCH_CHECK(Hq == Hkv, "attn_bwd_bsd: direct dv_out requires Hq == Hkv");
    TORCH_CHECK(dv_out->stride(3) == 1, "attn_bwd_bsd: dv_out last dim must be contiguous");
    dv_full = *dv_out;
    dv_sb = dv_full.stride(0);
    dv_sh = dv_full.stride(1);
    dv_sr = dv_full.stride(2);
    dv_ptr = dv_full.data_ptr();
  } else {
    dv_full = at::empty({B, Hq, S, D}, q.options());
    dv_ptr = dv_full.data_ptr();
  }
  DK_OK(dk_attn_bwd_dkdv(dk_full.data_ptr(), dv_ptr, do_bsd.data_ptr(),
                         q.data_ptr(), k.data_ptr(), v.data_ptr(), lse.data_ptr<float>(),
                         delta.data_ptr<float>(), B, Hq, Hkv, S, D, (float)scale,
                         sb, sh, sr, v_sb, v_sh, v_sr, dv_sb, dv_sh, dv_sr,
                         dt_of(q), stream()));
  DK_OK(dk_attn_bwd_dq(dq.data_ptr(), do_bsd.data_ptr(), q.data_ptr(), k.data_ptr(),
                       v.data_ptr(), lse.data_ptr<float>(), delta.data_ptr<float>(),
                       B, Hq, Hkv, S, D, (float)scale, sb, sh, sr, v_sb, v_sh, v_sr,
                       dt_of(q), stream()));
  return {dq, dk_full, dv_full};
}

// ---- split-K weight-gradient GEMM (dW = dy^T x) ----
// The dW GEMMs reduce over R = 32k tokens into small [N, K] outputs: a
// single GEMM yields only ~50-100 workgroups on a 256-CU chip (grid
// starvation — measured 471-951 TF vs 1300-1475 TF for the fwd/dx classes).
// Split the token dimension into `nchunk` slabs computed as ONE rocBLAS
// strided-batched bf16->fp32 GEMM (batch x tiles fills the chip), then
// dk_accum_chunks reduces the fp32 partials deterministically into the
// fp32 master gradient — numerically STRONGER than the single bf16-out
// GEMM (partials never round to bf16).  Replaces the reference's
// F.linear weight-grad + autocast cast (train_fsdp.py:383 grad flow).

static rocblas_handle dw_handle() {
  static rocblas_handle h = [] {
    rocblas_handle hh;
    TORCH_CHECK(rocblas_create_handle(&hh) == rocblas_status_success);
    return hh;
  }();
  return h;
}

// hipBLASLt variant of the batched dW GEMM: same math, but the solution is
// picked by timing the heuristic's top candidates once per shape (a
// mini-TunableOp; rocBLAS tops out ~1000-1055 TF on these batched
// bf16->fp32 shapes).  DK_DW_BLASLT=0 falls back to the rocBLAS path.


#define LT_OK(call) TORCH_CHECK((call) == HIPBLAS_STATUS_SUCCESS, #call " failed")

static hipblasLtHandle_t lt_handle() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t hh;
    LT_OK(hipblasLtCreate(&hh));
    return hh;
  }();
  return h;
}

static at::Tensor& lt_workspace() {
  static at::Tensor ws;  // lives for the process; torch caching allocator
  if (!ws.defined())
    ws = at::empty({64 * 1024 * 1024},
                   at::TensorOptions().dtype(at::kByte).device(at::kCUDA));
  return ws;
}

struct LtPlan {
  hipblasLtMatmulDesc_t op;
  hipblasLtMatrixLayout_t la, lb, lc;
  hipblasLtMatmulAlgo_t algo;
};

static bool dw_gemm_blaslt(const at::Tensor& dy, const at::Tensor& x,
                           at::Tensor& partials, int64_t Rc) {
  static std::map<std::tuple<int64_t, int64_t, int64_t, int64_t>, LtPlan> plans;
  const int64_t N = dy.size(1), K = x.size(1), nchunk = partials.size(0);
  const bool is_bf16 = dy.scalar_type() == at::kBFloat16;
  const hipDataType ab_t = is_bf16 ? HIP_R_16BF : HIP_R_16F;
  auto key = std::make_tuple(N, K, Rc, nchunk);
  auto it = plans.find(key);
  const float alpha = 1.f, beta = 0.f;
  hipStream_t strm = (hipStream_t)stream();
  at::Tensor& ws = lt_workspace();
  const size_t ws_size = (size_t)ws.numel();

  if (it == plans.end()) {
    LtPlan p;
    LT_OK(hipblasLtMatmulDescCreate(&p.op, HIPBLAS_COMPUTE_32F, HIP_R_32F));
    hipblasOperation_t ta = HIPBLAS_OP_N, tb = HIPBLAS_OP_T;
    LT_OK(hipblasLtMatmulDescSetAttribute(p.op, HIPBLASLT_MATMUL_DESC_TRANSA, &ta, sizeof(ta)));
    LT_OK(hipblasLtMatmulDescSetAttribute(p.op, HIPBLASLT_MATMUL_DESC_TRANSB, &tb, sizeof(tb)));
    // column-major: C[K, N] = A[K, Rc] (x chunk) * B[N, Rc]^T (dy chunk)
    auto mk = [&](hipblasLtMatrixLayout_t* l, hipDataType t, int64_t rows,
                  int64_t cols, int64_t ld, int64_t stride) {
      LT_OK(hipblasLtMatrixLayoutCreate(l, t, rows, cols, ld));
      int32_t bc = (int32_t)nchunk;
      LT_OK(hipblasLtMatrixLayoutSetAttribute(*l, HIPBLASLT_MATRIX_LAYOUT_BATCH_COUNT,
                                              &bc, sizeof(bc)));
      int64_t so = stride;
      LT_OK(hipblasLtMatrixLayoutSetAttribute(
          *l, HIPBLASLT_MATRIX_LAYOUT_STRIDED_BATCH_OFFSET, &so, sizeof(so)));
    };
    mk(&p.la, ab_t, K, Rc, K, Rc * K);
    mk(&p.lb, ab_t, N, Rc, N, Rc * N);
    mk(&p.lc, HIP_R_32F, K, N, K, N * K);

    hipblasLtMatmulPreference_t pref;
    LT_OK(hipblasLtMatmulPreferenceCreate(&pref));
    uint64_t mws = ws_size;
    LT_OK(hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &mws, sizeof(mws)));
    constexpr int kReq = 12;
    hipblasLtMatmulHeuristicResult_t res[kReq];
    int nres = 0;
    hipblasStatus_t hst = hipblasLtMatmulAlgoGetHeuristic(
        lt_handle(), p.op, p.la, p.lb, p.lc, p.lc, pref, kReq, res, &nres);
    hipblasLtMatmulPreferenceDestroy(pref);
    if (hst != HIPBLAS_STATUS_SUCCESS || nres == 0) return false;

    // time each candidate once (3 reps) and keep the fastest
    float best_ms = 1e30f;
    int best = -1;
    hipEvent_t e0, e1;
    hipEventCreate(&e0);
    hipEventCreate(&e1);
    for (int i = 0; i < nres; ++i) {
      auto run = [&] {
        return hipblasLtMatmul(lt_handle(), p.op, &alpha, x.data_ptr(), p.la,
                               dy.data_ptr(), p.lb, &beta, partials.data_ptr(), p.lc,
                               partials.data_ptr(), p.lc, &res[i].algo,
                               ws.data_ptr(), ws_size, strm);
      };
      if (run() != HIPBLAS_STATUS_SUCCESS) continue;
      hipEventRecord(e0, strm);
      for (int r = 0; r < 3; ++r) (void)run();
      hipEventRecord(e1, strm);
      hipEventSynchronize(e1);
      float ms = 0;
      hipEventElapsedTime(&ms, e0, e1);
      if (ms < best_ms) { best_ms = ms; best = i; }
    }
    hipEventDestroy(e0);
    hipEventDestroy(e1);
    if (best < 0) return false;
    p.algo = res[best].algo;
    it = plans.emplace(key, p).first;
  }
  const LtPlan& p = it->second;
  hipblasStatus_t st = hipblasLtMatmul(
      lt_handle(), p.op, &alpha, x.data_ptr(), p.la, dy.data_ptr(), p.lb, &beta,
      partials.data_ptr(), p.lc, partials.data_ptr(), p.lc, &p.algo,
      lt_workspace().data_ptr(), ws_size, strm);
  return st == HIPBLAS_STATUS_SUCCESS;
}

static bool dw_use_blaslt() {
  // default OFF: the hipBLASLt algo is picked by runtime timing, so the
  // solution (and the bitwise result) could vary run to run; the rocBLAS
  // path with the committed chunk table is deterministic and measured
  // equal in-step (927 vs 930 ms, tools/dw_sweep A/B).  DK_DW_BLASLT=1
  // opts in.
  static int v = [] {
    const char* e = getenv("DK_DW_BLASLT");
    return (e && e[0] == '1') ? 1 : 0;
  }();
  return v != 0;
}

void dw_gemm_batched(const at::Tensor& dy, const at::Tensor& x, at::Tensor& partials) {
  // dy [R, N], x [R, K] (bf16/f16, contiguous); partials [nchunk, N, K] fp32
  CHECK_DEV_CONTIG(dy);
  CHECK_DEV_CONTIG(x);
  CHECK_DEV_CONTIG(partials);
  TORCH_CHECK(partials.scalar_type() == at::kFloat);
  TORCH_CHECK(dy.scalar_type() == x.scalar_type());
  const int64_t R = dy.size(0), N = dy.size(1), K = x.size(1);
  const int64_t nchunk = partials.size(0);
  TORCH_CHECK(x.size(0) == R && partials.size(1) == N && partials.size(2) == K);
  TORCH_CHECK(R % nchunk == 0, "R must divide into nchunk");
  const int64_t Rc = R / nchunk;
  if (dw_use_blaslt() && dw_gemm_blaslt(dy, x, partials, Rc)) return;
  rocblas_handle h = dw_handle();
  TORCH_CHECK(rocblas_set_stream(h, (hipStream_t)stream()) == rocblas_status_success);
  const rocblas_datatype ab_t = dy.scalar_type() == at::kBFloat16
                                    ? rocblas_datatype_bf16_r : rocblas_datatype_f16_r;
  const float alpha = 1.f, beta = 0.f;
  // column-major mapping: D_cm[K, N] = x_chunk_cm[K, Rc] (N) * dy_chunk[Rc, N] (T)
  rocblas_status st = rocblas_gemm_strided_batched_ex(
      h, rocblas_operation_none, rocblas_operation_transpose,
      (rocblas_int)K, (rocblas_int)N, (rocblas_int)Rc, &alpha,
      x.data_ptr(), ab_t, (rocblas_int)K, Rc * K,
      dy.data_ptr(), ab_t, (rocblas_int)N, Rc * N, &beta,
      partials.data_ptr(), rocblas_datatype_f32_r, (rocblas_int)K, N * K,
      partials.data_ptr(), rocblas_datatype_f32_r, (rocblas_int)K, N * K,
      (rocblas_int)nchunk, rocblas_datatype_f32_r, rocblas_gemm_algo_standard,
      0, 0);
  TORCH_CHECK(st == rocblas_status_success, "rocblas_gemm_strided_batched_ex: ", (int)st);
}

void accum_chunks_(at::Tensor& dst, const at::Tensor& partials, int64_t elem_offset) {
  // dst (fp32, n elems) += sum over partials[b][elem_offset : elem_offset+n]
  CHECK_DEV_CONTIG(dst);
  CHECK_DEV_CONTIG(partials);
  TORCH_CHECK(dst.scalar_type() == at::kFloat && partials.scalar_type() == at::kFloat);
  const int64_t nchunk = partials.size(0);
  const int64_t chunk_stride = partials.numel() / nchunk;
  TORCH_CHECK(elem_offset + dst.numel() <= chunk_stride);
  DK_OK(dk_accum_chunks(dst.data_ptr<float>(),
                        partials.data_ptr<float>() + elem_offset,
                        dst.numel(), (int)nchunk, chunk_stride, stream()));
}

// fp32 master-grad += low-precision dW (fused cast+accumulate)
void accum_(at::Tensor& dst, const at::Tensor& src) {
  CHECK_DEV_CONTIG(dst);
  CHECK_DEV_CONTIG(src);
  TORCH_CHECK(dst.scalar_type() == at::kFloat, "accum_: dst must be fp32");
  TORCH_CHECK(dst.numel() == src.numel(), "accum_: numel mismatch");
  DK_OK(dk_accum(dst.data_ptr<float>(), src.data_ptr(), dst.numel(),
                 dt_of(src), stream()));
}

// ---- optimizer / outer step ----
void fused_adamw(at::Tensor& p, const at::Tensor& g, at::Tensor& m, at::Tensor& v,
                 double lr, double beta1, double beta2, double eps, double weight_decay,
                 int64_t step) {
  CHECK_DEV_CONTIG(p);
  DK_OK(dk_fused_adamw(p.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr<float>(),
                       v.data_ptr<float>(), p.numel(), (float)lr, (float)beta1,
                       (float)beta2, (float)eps, (float)weight_decay, (int)step, stream()));
}

at::Tensor clip_grad_(at::Tensor& g, double max_norm) {
  CHECK_DEV_CONTIG(g);
  const int grid = dk_gradsq_grid(g.numel());
  auto partials = at::empty({grid}, g.options().dtype(at::kFloat));
  auto out2 = at::empty({2}, g.options().dtype(at::kFloat));
  DK_OK(dk_grad_sq_partials(partials.data_ptr<float>(), g.data_ptr<float>(), g.numel(),
                            grid, stream()));
  DK_OK(dk_clip_apply(g.data_ptr<float>(), out2.data_ptr<float>(), partials.data_ptr<float>(),
                      grid, g.numel(), (float)max_norm, stream()));
  return out2;  // [total_norm, applied_coef]
}

at::Tensor grad_norm(const at::Tensor& g) {
  // norm only (no scaling): reuse partials + finalize with max_norm=inf
  const int grid = dk_gradsq_grid(g.numel());
  auto partials = at::empty({grid}, g.options().dtype(at::kFloat));
  DK_OK(dk_grad_sq_partials(partials.data_ptr<float>(), g.data_ptr<float>(), g.numel(),
                            grid, stream()));
  return partials.sum().sqrt();
}

void pseudo_grad(at::Tensor& g_out, const at::Tensor& theta_outer,
                 const at::Tensor& theta_local) {
  DK_OK(dk_pseudo_grad(g_out.data_ptr<float>(), theta_outer.data_ptr<float>(),
                       theta_local.data_ptr<float>(), g_out.numel(), stream()));
}

void outer_nesterov(at::Tensor& theta_outer, at::Tensor& theta_local, at::Tensor& buf,
                    const at::Tensor& g_avg, double lr, double momentum, bool first) {
  DK_OK(dk_outer_nesterov(theta_outer.data_ptr<float>(), theta_local.data_ptr<float>(),
                          buf.data_ptr<float>(), g_avg.data_ptr<float>(),
                          theta_outer.numel(), (float)lr, (float)momentum, first ? 1 : 0,
                          stream()));
}

void cast_(at::Tensor& dst, const at::Tensor& src) {
  TORCH_CHECK(dst.numel() == src.numel());
  DK_OK(dk_cast(dst.data_ptr(), src.data_ptr(), dst.numel(), dt_of(dst), dt_of(src), stream()));
}

// ---- probe (test-only) ----
at::Tensor probe_mfma(const at::Tensor& a, const at::Tensor& b) {
  TORCH_CHECK(a.scalar_type() == at::kBFloat16 && a.numel() == 16 * 32);
  TORCH_CHECK(b.scalar_type() == at::kBFloat16 && b.numel() == 32 * 16);
  auto out = at::zeros({16, 16}, a.options().dtype(at::kFloat));
  DK_OK(dk_probe_mfma_16x16x32_bf16(out.data_ptr<float>(), a.data_ptr(), b.data_ptr(),
                                    stream()));
  return out;
}

at::Tensor probe_mfma_alt(const at::Tensor& a, const at::Tensor& b) {
  auto out = at::zeros({16, 16}, a.options().dtype(at::kFloat));
  DK_OK(dk_probe_mfma_16x16x32_bf16_alt(out.data_ptr<float>(), a.data_ptr(), b.data_ptr(),
                                        stream()));
  return out;
}

at::Tensor probe_mfma32(const at::Tensor& a, const at::Tensor& b) {
  TORCH_CHECK(a.numel() == 32 * 16 && b.numel() == 16 * 32);
  auto out = at::zeros({32, 32}, a.options().dtype(at::kFloat));
  DK_OK(dk_probe_mfma_32x32x16_bf16(out.data_ptr<float>(), a.data_ptr(), b.data_ptr(),
                                    stream()));
  return out;
}

at::Tensor probe_permlane32(const at::Tensor& ref_dev) {
  auto out = at::zeros({2, 64}, ref_dev.options().dtype(at::kInt));
  DK_OK(dk_probe_permlane32(out.data_ptr<int>(), stream()));
  return out;
}

std::string version() { return dk_version(); }

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("rmsnorm_add_fwd", &rmsnorm_add_fwd);
  m.def("rope", &rope);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("swiglu2_fwd", &swiglu2_fwd);
  m.def("swiglu2_bwd", &swiglu2_bwd);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("qkv_rope_gather", &qkv_rope_gather);
  m.def("rope_scatter_", &rope_scatter_);
  m.def("attn_fwd_bsd", &attn_fwd_bsd);
  m.def("accum_", &accum_);
  m.def("dw_gemm_batched", &dw_gemm_batched);
  m.def("accum_chunks_", &accum_chunks_);
  m.def("attn_bwd_bsd", &attn_bwd_bsd, py::arg("do_bsd"), py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("o_bsd"), py::arg("lse"), py::arg("scale"),
        py::arg("dv_out") = py::none());
  m.def("fused_adamw", &fused_adamw);
  m.def("clip_grad_", &clip_grad_);
  m.def("grad_norm", &grad_norm);
  m.def("pseudo_grad", &pseudo_grad);
  m.def("outer_nesterov", &outer_nesterov);
  m.def("cast_", &cast_);
  m.def("probe_mfma", &probe_mfma);
  m.def("probe_mfma_alt", &probe_mfma_alt);
  m.def("probe_mfma32", &probe_mfma32);
  m.def("probe_permlane32", &probe_permlane32);
  m.def("version", &version);
}
