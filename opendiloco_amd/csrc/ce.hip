// ce.hip — fused causal-LM cross-entropy for gfx950.
//
// Replaces the loss computation inside transformers LlamaForCausalLM
// (logits -> fp32, shifted CE with mean reduction) that the reference hits
// via model(**batch) at train_fsdp.py:378 / train_diloco_torch.py:313.
// The causal SHIFT lives inside the kernels: the caller passes the FULL
// [B, S, V] logits and [B, S] labels; row (b, s<S-1) scores logits[b,s,:]
// against labels[b,s+1] (T = B*(S-1) rows in the mean), and the backward
// writes the full [B, S, V] gradient with the last position zeroed — no
// 2.1 GB slice copy / pad round trips.  Forward is a single online-softmax
// pass; backward one read + one write.  No atomics — deterministic.

#include "dk_common.h"
#include "../../include/diloco_kernels.h"

#include <math.h>

__device__ __forceinline__ void merge_ms(float& m, float& s, float om, float os) {
  float nm = fmaxf(m, om);
  float t1 = (m == -INFINITY) ? 0.f : s * __expf(m - nm);
  float t2 = (om == -INFINITY) ? 0.f : os * __expf(om - nm);
  m = nm;
  s = t1 + t2;
}

template <int DT, int NT>
__global__ void ce_fwd_kernel(float* __restrict__ loss_rows, float* __restrict__ lse_out,
                              const typename DTraits<DT>::T* __restrict__ logits,
                              const int64_t* __restrict__ labels, int64_t B, int64_t Sm1,
                              int64_t S, int64_t V) {
  using TR = DTraits<DT>;
  using TT = typename TR::T;
  constexpr int W = VecIO<TT>::W;
  using V8 = typename VecIO<TT>::V;
  __shared__ float sm[NT / DK_WAVE], ss[NT / DK_WAVE];

  const int64_t nvec = V / W;
  const int64_t T = B * Sm1;
  for (int64_t r = blockIdx.x; r < T; r += gridDim.x) {
    const int64_t b = r / Sm1, sp = r % Sm1;   // scores pos sp vs label sp+1
    const TT* row = logits + (b * S + sp) * V;
    const int64_t lab_r = labels[b * S + sp + 1];
    float m = -INFINITY, s = 0.f;
    for (int64_t i = threadIdx.x; i < nvec; i += NT) {
      V8 xv = *(const V8*)(row + i * W);
      // per-vector max first, then one rescale + W independent exps: breaks
      // the per-element serial rescale chain (latency-bound) into ILP
      float xf[W], vm = -INFINITY;
      packed_to_f32<DT, W>(&xv, xf);
#pragma unroll
      for (int j = 0; j < W; ++j) vm = fmaxf(vm, xf[j]);
      if (vm > m) { s *= __expf(m - vm); m = vm; }
      float ps = 0.f;
#pragma unroll
      for (int j = 0; j < W; ++j) ps += __expf(xf[j] - m);
      s += ps;
    }
    for (int64_t c = nvec * W + threadIdx.x; c < V; c += NT) {
      float x = TR::toF(row[c]);
      if (x > m) { s *= __expf(m - x); m = x; }
      s += __expf(x - m);
    }
    // wave merge
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float om = __shfl_xor(m, off, DK_WAVE);
      float os = __shfl_xor(s, off, DK_WAVE);
      merge_ms(m, s, om, os);
    }
    const int wid = threadIdx.x / DK_WAVE;
    if ((threadIdx.x & (DK_WAVE - 1)) == 0) { sm[wid] = m; ss[wid] = s; }
    __syncthreads();
    if (threadIdx.x == 0) {
#pragma unroll
      for (int i = 1; i < NT / DK_WAVE; ++i) merge_ms(m, s, sm[i], ss[i]);
      float lse = m + __logf(s);
      lse_out[r] = lse;
      loss_rows[r] = lse - TR::toF(row[lab_r]);
    }
    __syncthreads();
  }
}

template <int DT, int NT>
__global__ void ce_bwd_kernel(typename DTraits<DT>::T* __restrict__ dlogits,
                              const typename DTraits<DT>::T* __restrict__ logits,
                              const float* __restrict__ lse,
                              const int64_t* __restrict__ labels,
                              const float* __restrict__ dloss, float inv_T,
                              int64_t B, int64_t Sm1, int64_t S, int64_t V) {
  using TR = DTraits<DT>;
  using TT = typename TR::T;
  constexpr int W = VecIO<TT>::W;
  using V8 = typename VecIO<TT>::V;
  const float scale = dloss[0] * inv_T;
  const int64_t nvec = V / W;
  const int64_t rows_all = B * S;  // covers the zeroed last position too
  for (int64_t rr = blockIdx.x; rr < rows_all; rr += gridDim.x) {
    const int64_t b = rr / S, sp = rr % S;
    TT* drow0 = dlogits + rr * V;
    if (sp == S - 1) {  // no next-token target: zero gradient row
      for (int64_t i = threadIdx.x; i < nvec; i += NT) {
        V8 z = (V8)(0);
        *(V8*)(drow0 + i * W) = z;
      }
      for (int64_t c = nvec * W + threadIdx.x; c < V; c += NT) drow0[c] = TR::fromF(0.f);
      continue;
    }
    const int64_t r = b * Sm1 + sp;
    const TT* row = logits + rr * V;
    TT* drow = drow0;
    const float l = lse[r];
    const int64_t lab = labels[rr + 1];
    for (int64_t i = threadIdx.x; i < nvec; i += NT) {
      V8 xv = *(const V8*)(row + i * W);
      V8 dv;
      float xf[W];
      packed_to_f32<DT, W>(&xv, xf);
#pragma unroll
      for (int j = 0; j < W; ++j) {
        int64_t v = i * W + j;
        float p = __expf(xf[j] - l);
        float d = (p - (v == lab ? 1.f : 0.f)) * scale;
        ((TT*)&dv)[j] = TR::fromF(d);
      }
      *(V8*)(drow + i * W) = dv;
    }
    for (int64_t c = nvec * W + threadIdx.x; c < V; c += NT) {
      float p = __expf(TR::toF(row[c]) - l);
      drow[c] = TR::fromF((p - (c == lab ? 1.f : 0.f)) * scale);
    }
  }
}

extern "C" int dk_cross_entropy_fwd(float* loss_rows, float* lse, const void* logits,
                                    const int64_t* labels, int64_t B, int64_t S,
                                    int64_t V, int dtype, dkStream stream) {
  constexpr int NT = 256;
  const int64_t T = B * (S - 1);
  int grid = (int)(T < 2048 ? T : 2048);
  DK_DISPATCH_DT(dtype, {
    using TT = typename DTraits<kDT>::T;
    hipLaunchKernelGGL((ce_fwd_kernel<kDT, NT>), dim3(grid), dim3(NT), 0,
                       (hipStream_t)stream, loss_rows, lse, (const TT*)logits, labels,
                       B, S - 1, S, V);
  });
  DK_CHECK_LAUNCH();
  return 0;
}

extern "C" int dk_cross_entropy_bwd(void* dlogits, const void* logits, const float* lse,
                                    const int64_t* labels, const float* dloss,
                                    float inv_T, int64_t B, int64_t S, int64_t V,
                                    int dtype, dkStream stream) {
  constexpr int NT = 256;
  const int64_t rows_all = B * S;
  int grid = (int)(rows_all < 2048 ? rows_all : 2048);
  DK_DISPATCH_DT(dtype, {
    using TT = typename DTraits<kDT>::T;
    hipLaunchKernelGGL((ce_bwd_kernel<kDT, NT>), dim3(grid), dim3(NT), 0,
                       (hipStream_t)stream, (TT*)dlogits, (const TT*)logits, lse, labels,
                       dloss, inv_T, B, S - 1, S, V);
  });
  DK_CHECK_LAUNCH();
  return 0;
}
