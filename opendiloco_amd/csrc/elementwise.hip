// elementwise.hip — memory-bound kernels of the DiLoCo hot path (gfx950).
//
// RMSNorm fwd/bwd, RoPE, SwiGLU, fused AdamW, grad-clip, pseudo-gradient,
// outer Nesterov SGD, casts.  All HBM-bound: the design rules applied are
// vectorized 16-B loads (guide G13), grid-stride loops capped at ~2048 blocks
// (G11), wave-shuffle reductions (no serial lanes), and deterministic
// two-pass reductions (no atomics) so loss traces are reproducible run-to-run
// (the reference's tests compare per-step losses at atol 1e-3,
// tests/test_training/test_train.py:82).

#include "dk_common.h"
#include "../../include/diloco_kernels.h"

#include <math.h>

// ====================== RMSNorm ======================
// forward: y = w * (x * rsqrt(mean(x^2)+eps)); matches transformers
// LlamaRMSNorm (fp32 internal, output rounded to input dtype).

// RES: fuse the residual add h = x + res (h written back, rounded to T
// exactly like a separate torch add would) into the normalisation pass.
template <int DT, int NT, bool RES>
__global__ void rmsnorm_fwd_kernel(typename DTraits<DT>::T* __restrict__ y,
                                   typename DTraits<DT>::T* __restrict__ h_out,
                                   float* __restrict__ invrms,
                                   const typename DTraits<DT>::T* __restrict__ x,
                                   const typename DTraits<DT>::T* __restrict__ res,
                                   const typename DTraits<DT>::T* __restrict__ w,
                                   int64_t rows, int cols, float eps) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  constexpr int W = VecIO<T>::W;
  using V = typename VecIO<T>::V;
  __shared__ float sred[NT / DK_WAVE];

  const int nvec = cols / W;
  for (int64_t r = blockIdx.x; r < rows; r += gridDim.x) {
    const T* xr = (RES ? h_out : x) + r * cols;  // second pass reads h
    T* yr = y + r * cols;
    float ss = 0.f;
    if (RES) {
      const T* xin = x + r * cols;
      const T* rin = res + r * cols;
      T* hr = h_out + r * cols;
      for (int i = threadIdx.x; i < nvec; i += NT) {
        V xv = *(const V*)(xin + i * W);
        V rv = *(const V*)(rin + i * W);
        V hv;
#pragma unroll
        for (int j = 0; j < W; ++j) {
          T h = TR::fromF(TR::toF(((const T*)&xv)[j]) + TR::toF(((const T*)&rv)[j]));
          ((T*)&hv)[j] = h;
          float f = TR::toF(h);
          ss += f * f;
        }
        *(V*)(hr + i * W) = hv;
      }
      for (int c = nvec * W + threadIdx.x; c < cols; c += NT) {
        T h = TR::fromF(TR::toF(xin[c]) + TR::toF(rin[c]));
        hr[c] = h;
        float f = TR::toF(h);
        ss += f * f;
      }
    } else {
      for (int i = threadIdx.x; i < nvec; i += NT) {
        V xv = *(const V*)(xr + i * W);
#pragma unroll
        for (int j = 0; j < W; ++j) {
          float f = TR::toF(((const T*)&xv)[j]);
          ss += f * f;
        }
      }
      for (int c = nvec * W + threadIdx.x; c < cols; c += NT) {
        float f = TR::toF(xr[c]);
        ss += f * f;
      }
    }
    if (RES) __syncthreads();  // h_out writes visible to this block's own
                               // second pass (same thread re-reads its own
                               // writes; barrier orders sred reuse anyway)
    ss = block_reduce_sum<NT>(ss, sred);
    const float ir = rsqrtf(ss / (float)cols + eps);
    if (threadIdx.x == 0 && invrms) invrms[r] = ir;
    for (int i = threadIdx.x; i < nvec; i += NT) {
      V xv = *(const V*)(xr + i * W);
      V wv = *(const V*)(w + i * W);
      V yv;
#pragma unroll
      for (int j = 0; j < W; ++j) {
        // mirror HF order: xhat rounded to T first, then multiplied by w
        float xh = TR::toF(TR::fromF(TR::toF(((const T*)&xv)[j]) * ir));
        ((T*)&yv)[j] = TR::fromF(TR::toF(((const T*)&wv)[j]) * xh);
      }
      *(V*)(yr + i * W) = yv;
    }
    for (int c = nvec * W + threadIdx.x; c < cols; c += NT) {
      float xh = TR::toF(TR::fromF(TR::toF(xr[c]) * ir));
      yr[c] = TR::fromF(TR::toF(w[c]) * xh);
    }
  }
}

// vectorized forward for the hot hidden sizes (cols == NT*VEC): thread t
// owns columns [t*VEC, t*VEC+VEC) so the whole row lives in registers
// between the sum-of-squares pass and the normalize pass — x/res are read
// once, h/y written once, no strided second read (fwd 4.2 -> ~6 TB/s).
template <int DT, int NT, int VEC, bool RES>
__global__ void rmsnorm_fwd_vec_kernel(typename DTraits<DT>::T* __restrict__ y,
                                       typename DTraits<DT>::T* __restrict__ h_out,
                                       float* __restrict__ invrms,
                                       const typename DTraits<DT>::T* __restrict__ x,
                                       const typename DTraits<DT>::T* __restrict__ res,
                                       const typename DTraits<DT>::T* __restrict__ w,
                                       int64_t rows, int cols, float eps) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  typedef __attribute__((ext_vector_type(VEC))) short vec_t;
  __shared__ float sred[NT / DK_WAVE];
  const int c0 = threadIdx.x * VEC;
  float wf[VEC];
  {
    vec_t wvv = *(const vec_t*)(w + c0);
    packed_to_f32<DT, VEC>(&wvv, wf);
  }
  // software pipeline: next row's loads are issued before this row's
  // block-reduce barrier so HBM latency overlaps the sync
  vec_t xv, rv;
  if (blockIdx.x < rows) {
    xv = *(const vec_t*)(x + blockIdx.x * cols + c0);
    if (RES) rv = *(const vec_t*)(res + blockIdx.x * cols + c0);
  }
  for (int64_t r = blockIdx.x; r < rows; r += gridDim.x) {
    float hf[VEC], ss = 0.f;
    if (RES) {
      vec_t hv;
      float xf[VEC], rf[VEC];
      packed_to_f32<DT, VEC>(&xv, xf);
      packed_to_f32<DT, VEC>(&rv, rf);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        T h = TR::fromF(xf[j] + rf[j]);
        ((T*)&hv)[j] = h;
        hf[j] = TR::toF(h);
        ss += hf[j] * hf[j];
      }
      *(vec_t*)(h_out + r * cols + c0) = hv;
    } else {
      packed_to_f32<DT, VEC>(&xv, hf);
#pragma unroll
      for (int j = 0; j < VEC; ++j) ss += hf[j] * hf[j];
    }
    if (r + gridDim.x < rows) {
      xv = *(const vec_t*)(x + (r + gridDim.x) * cols + c0);
      if (RES) rv = *(const vec_t*)(res + (r + gridDim.x) * cols + c0);
    }
    ss = block_reduce_sum<NT>(ss, sred);
    const float ir = rsqrtf(ss / (float)cols + eps);
    if (threadIdx.x == 0 && invrms) invrms[r] = ir;
    vec_t yv;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      // mirror HF order: xhat rounded to T first, then multiplied by w
      float xh = TR::toF(TR::fromF(hf[j] * ir));
      ((T*)&yv)[j] = TR::fromF(wf[j] * xh);
    }
    *(vec_t*)(y + r * cols + c0) = yv;
  }
}

// backward: let xhat = x*invrms, g = dy*w.
//   dx = (g - xhat * mean(g*xhat)) * invrms
//   dw partials accumulate in REGISTERS per block (thread t owns columns
//   t, t+NT, ...) and are written once at the end — no global RMW, no
//   zero-init, deterministic (block b owns rows b, b+grid, ...).
// MAXC = max columns per thread: cols <= NT*MAXC (8192 at NT=256).
template <int DT, int NT, int MAXC, bool DRES>
__global__ void rmsnorm_bwd_kernel(typename DTraits<DT>::T* __restrict__ dx,
                                   float* __restrict__ dwp,
                                   const typename DTraits<DT>::T* __restrict__ dy,
                                   const typename DTraits<DT>::T* __restrict__ dres,
                                   const typename DTraits<DT>::T* __restrict__ x,
                                   const typename DTraits<DT>::T* __restrict__ w,
                                   const float* __restrict__ invrms,
                                   int64_t rows, int cols) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  __shared__ float sred[NT / DK_WAVE];
  float dwacc[MAXC];
#pragma unroll
  for (int k = 0; k < MAXC; ++k) dwacc[k] = 0.f;

  for (int64_t r = blockIdx.x; r < rows; r += gridDim.x) {
    const T* xr = x + r * cols;
    const T* dyr = dy + r * cols;
    T* dxr = dx + r * cols;
    const float ir = invrms[r];
    float dot = 0.f;
    for (int c = threadIdx.x; c < cols; c += NT) {
      float xh = TR::toF(xr[c]) * ir;
      float g = TR::toF(dyr[c]) * TR::toF(w[c]);
      dot += g * xh;
    }
    dot = block_reduce_sum<NT>(dot, sred) / (float)cols;
    const T* drr = DRES ? dres + r * cols : nullptr;
    int k = 0;
    for (int c = threadIdx.x; c < cols; c += NT, ++k) {
      float xh = TR::toF(xr[c]) * ir;
      float dyf = TR::toF(dyr[c]);
      float g = dyf * TR::toF(w[c]);
      float dxv = (g - xh * dot) * ir;
      if (DRES) dxv += TR::toF(drr[c]);
      dxr[c] = TR::fromF(dxv);
      dwacc[k] += dyf * xh;
    }
  }
  {
    int k = 0;
    float* dwrow = dwp + (int64_t)blockIdx.x * cols;
    for (int c = threadIdx.x; c < cols; c += NT, ++k) dwrow[c] = dwacc[k];
  }
}

// two-stage deterministic reduce of [grid][cols] -> [cols]:
//  stage 1: block (y, cc) sums a 32-row slice -> tmp[y][cols]  (coalesced)
//  stage 2: sums the <=64 tmp rows -> out[cols]
__global__ void reduce_partials_stage1(float* __restrict__ tmp,
                                       const float* __restrict__ partial,
                                       int grid, int nslices, int64_t cols) {
  const int y = blockIdx.y;
  const int g0 = y * 32;
  const int g1 = min(grid, g0 + 32);
  for (int64_t c = blockIdx.x * blockDim.x + threadIdx.x; c < cols;
       c += (int64_t)gridDim.x * blockDim.x) {
    float s = 0.f;
    for (int g = g0; g < g1; ++g) s += partial[(int64_t)g * cols + c];
    tmp[(int64_t)y * cols + c] = s;
  }
}

__global__ void reduce_partials_stage2(float* __restrict__ out,
                                       const float* __restrict__ tmp,
                                       int nslices, int64_t cols) {
  for (int64_t c = blockIdx.x * blockDim.x + threadIdx.x; c < cols;
       c += (int64_t)gridDim.x * blockDim.x) {
    float s = 0.f;
    for (int g = 0; g < nslices; ++g) s += tmp[(int64_t)g * cols + c];
    out[c] = s;
  }
}

// vectorized rmsnorm backward for cols == NT*VEC (the 150m/1b hidden sizes):
// thread t owns the contiguous columns [t*VEC, (t+1)*VEC); w is loaded into
// registers ONCE per block; each row is read once (values kept in registers
// between the dot pass and the dx/dw pass) with shortx4/8 vector loads.
template <int DT, int NT, int VEC, bool DRES>
__global__ void rmsnorm_bwd_vec_kernel(typename DTraits<DT>::T* __restrict__ dx,
                                       float* __restrict__ dwp,
                                       const typename DTraits<DT>::T* __restrict__ dy,
                                       const typename DTraits<DT>::T* __restrict__ dres,
                                       const typename DTraits<DT>::T* __restrict__ x,
                                       const typename DTraits<DT>::T* __restrict__ w,
                                       const float* __restrict__ invrms,
                                       int64_t rows, int cols) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  typedef __attribute__((ext_vector_type(VEC))) short vec_t;
  __shared__ float sred[2 * (NT / DK_WAVE)];
  const int c0 = threadIdx.x * VEC;
  float wv[VEC], dwacc[VEC];
  {
    vec_t wvv = *(const vec_t*)(w + c0);
    packed_to_f32<DT, VEC>(&wvv, wv);
#pragma unroll
    for (int j = 0; j < VEC; ++j) dwacc[j] = 0.f;
  }
  // TWO rows per iteration through ONE block-reduce barrier: the per-row
  // sync was the limiter (bwd 3.2 TB/s vs fwd 5.7 at one row per barrier);
  // bitwise identical to the 1-row loop (each row keeps its own wave-sum
  // order, dw accumulates rows in the same block sequence)
  const int64_t G = gridDim.x;
  vec_t xv0, dv0, rv0, xv1, dv1, rv1;
  auto ld = [&](int64_t r, vec_t& xvv, vec_t& dvv, vec_t& rvv) {
    xvv = *(const vec_t*)(x + r * cols + c0);
    dvv = *(const vec_t*)(dy + r * cols + c0);
    if (DRES) rvv = *(const vec_t*)(dres + r * cols + c0);
  };
  if (blockIdx.x < rows) ld(blockIdx.x, xv0, dv0, rv0);
  if (blockIdx.x + G < rows) ld(blockIdx.x + G, xv1, dv1, rv1);
  for (int64_t r = blockIdx.x; r < rows; r += 2 * G) {
    const int64_t r1 = r + G;
    const bool has1 = r1 < rows;
    const float ir0 = invrms[r];
    const float ir1 = has1 ? invrms[r1] : 0.f;
    float xf0[VEC], dyf0[VEC], drf0[VEC], xf1[VEC], dyf1[VEC], drf1[VEC];
    packed_to_f32<DT, VEC>(&xv0, xf0);
    packed_to_f32<DT, VEC>(&dv0, dyf0);
    if (DRES) packed_to_f32<DT, VEC>(&rv0, drf0);
    if (has1) {
      packed_to_f32<DT, VEC>(&xv1, xf1);
      packed_to_f32<DT, VEC>(&dv1, dyf1);
      if (DRES) packed_to_f32<DT, VEC>(&rv1, drf1);
    }
    if (r + 2 * G < rows) ld(r + 2 * G, xv0, dv0, rv0);
    if (r1 + 2 * G < rows) ld(r1 + 2 * G, xv1, dv1, rv1);
    float s0 = 0.f, s1 = 0.f;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      s0 += dyf0[j] * wv[j] * xf0[j];
      if (has1) s1 += dyf1[j] * wv[j] * xf1[j];
    }
    block_reduce_sum2<NT>(s0, s1, sred);
    const float dot0 = s0 * ir0 / (float)cols;
    const float dot1 = s1 * ir1 / (float)cols;
    vec_t dxv0, dxv1;
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float xh = xf0[j] * ir0;
      float g = dyf0[j] * wv[j];
      float dvv = (g - xh * dot0) * ir0;
      if (DRES) dvv += drf0[j];
      ((T*)&dxv0)[j] = TR::fromF(dvv);
      dwacc[j] += dyf0[j] * xh;
    }
    *(vec_t*)(dx + r * cols + c0) = dxv0;
    if (has1) {
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float xh = xf1[j] * ir1;
        float g = dyf1[j] * wv[j];
        float dvv = (g - xh * dot1) * ir1;
        if (DRES) dvv += drf1[j];
        ((T*)&dxv1)[j] = TR::fromF(dvv);
        dwacc[j] += dyf1[j] * xh;
      }
      *(vec_t*)(dx + r1 * cols + c0) = dxv1;
    }
  }
  {
    float* dwrow = dwp + (int64_t)blockIdx.x * cols;
#pragma unroll
    for (int j = 0; j < VEC; ++j) dwrow[c0 + j] = dwacc[j];
  }
}

extern "C" int dk_rmsnorm_bwd_grid(int64_t rows) { return (int)(rows < 2048 ? rows : 2048); }

extern "C" int dk_rmsnorm_fwd(void* y, void* h_out, float* invrms, const void* x,
                              const void* res, const void* w, int64_t rows,
                              int64_t cols, float eps, int dtype, dkStream stream) {
  constexpr int NT = 256;
  int grid = (int)(rows < 4096 ? rows : 4096);
  DK_DISPATCH_DT(dtype, {
    using T = typename DTraits<kDT>::T;
    if constexpr (kDT != 0) {
      if (cols == NT * 4 || cols == NT * 8) {  // vectorized: row in registers
        if (cols == NT * 4) {
          if (res != nullptr)
            hipLaunchKernelGGL((rmsnorm_fwd_vec_kernel<kDT, NT, 4, true>), dim3(grid), dim3(NT), 0,
                               (hipStream_t)stream, (T*)y, (T*)h_out, invrms, (const T*)x,
                               (const T*)res, (const T*)w, rows, (int)cols, eps);
          else
            hipLaunchKernelGGL((rmsnorm_fwd_vec_kernel<kDT, NT, 4, false>), dim3(grid), dim3(NT), 0,
                               (hipStream_t)stream, (T*)y, nullptr, invrms, (const T*)x,
                               nullptr, (const T*)w, rows, (int)cols, eps);
        } else {
          if (res != nullptr)
            hipLaunchKernelGGL((rmsnorm_fwd_vec_kernel<kDT, NT, 8, true>), dim3(grid), dim3(NT), 0,
                               (hipStream_t)stream, (T*)y, (T*)h_out, invrms, (const T*)x,
                               (const T*)res, (const T*)w, rows, (int)cols, eps);
          else
            hipLaunchKernelGGL((rmsnorm_fwd_vec_kernel<kDT, NT, 8, false>), dim3(grid), dim3(NT), 0,
                               (hipStream_t)stream, (T*)y, nullptr, invrms, (const T*)x,
                               nullptr, (const T*)w, rows, (int)cols, eps);
        }
        hipError_t e = hipGetLastError();
        return (int)e;
      }
    }
    if (res != nullptr)
      hipLaunchKernelGGL((rmsnorm_fwd_kernel<kDT, NT, true>), dim3(grid), dim3(NT), 0,
                         (hipStream_t)stream, (T*)y, (T*)h_out, invrms, (const T*)x,
                         (const T*)res, (const T*)w, rows, (int)cols, eps);
    else
      hipLaunchKernelGGL((rmsnorm_fwd_kernel<kDT, NT, false>), dim3(grid), dim3(NT), 0,
                         (hipStream_t)stream, (T*)y, nullptr, invrms, (const T*)x,
                         nullptr, (const T*)w, rows, (int)cols, eps);
  });
  DK_CHECK_LAUNCH();
  return 0;
}

extern "C" int dk_rmsnorm_bwd(void* dx, float* dw_partial, const void* dy,
                              const void* dres, const void* x, const void* w,
                              const float* invrms, int64_t rows, int64_t cols,
                              int grid, int dtype, dkStream stream) {
  constexpr int NT = 256;
  if (cols > NT * 32) return (int)hipErrorInvalidValue;
  DK_DISPATCH_DT(dtype, {
    using T = typename DTraits<kDT>::T;
    if constexpr (kDT != 0) {
      // vectorized path for the hot hidden sizes
      if (cols == NT * 4) {
        if (dres != nullptr)
          hipLaunchKernelGGL((rmsnorm_bwd_vec_kernel<kDT, NT, 4, true>), dim3(grid), dim3(NT), 0,
                             (hipStream_t)stream, (T*)dx, dw_partial, (const T*)dy,
                             (const T*)dres, (const T*)x, (const T*)w, invrms, rows, (int)cols);
        else
          hipLaunchKernelGGL((rmsnorm_bwd_vec_kernel<kDT, NT, 4, false>), dim3(grid), dim3(NT), 0,
                             (hipStream_t)stream, (T*)dx, dw_partial, (const T*)dy,
                             nullptr, (const T*)x, (const T*)w, invrms, rows, (int)cols);
        hipError_t e = hipGetLastError();
        return (int)e;
      }
      if (cols == NT * 8) {
        if (dres != nullptr)
          hipLaunchKernelGGL((rmsnorm_bwd_vec_kernel<kDT, NT, 8, true>), dim3(grid), dim3(NT), 0,
                             (hipStream_t)stream, (T*)dx, dw_partial, (const T*)dy,
                             (const T*)dres, (const T*)x, (const T*)w, invrms, rows, (int)cols);
        else
          hipLaunchKernelGGL((rmsnorm_bwd_vec_kernel<kDT, NT, 8, false>), dim3(grid), dim3(NT), 0,
                             (hipStream_t)stream, (T*)dx, dw_partial, (const T*)dy,
                             nullptr, (const T*)x, (const T*)w, invrms, rows, (int)cols);
        hipError_t e = hipGetLastError();
        return (int)e;
      }
    }
    if (cols <= NT * 8) {
      if (dres != nullptr)
        hipLaunchKernelGGL((rmsnorm_bwd_kernel<kDT, NT, 8, true>), dim3(grid), dim3(NT), 0,
                           (hipStream_t)stream, (T*)dx, dw_partial, (const T*)dy,
                           (const T*)dres, (const T*)x, (const T*)w, invrms, rows, (int)cols);
      else
        hipLaunchKernelGGL((rmsnorm_bwd_kernel<kDT, NT, 8, false>), dim3(grid), dim3(NT), 0,
                           (hipStream_t)stream, (T*)dx, dw_partial, (const T*)dy,
                           nullptr, (const T*)x, (const T*)w, invrms, rows, (int)cols);
    } else {
      if (dres != nullptr)
        hipLaunchKernelGGL((rmsnorm_bwd_kernel<kDT, NT, 32, true>), dim3(grid), dim3(NT), 0,
                           (hipStream_t)stream, (T*)dx, dw_partial, (const T*)dy,
                           (const T*)dres, (const T*)x, (const T*)w, invrms, rows, (int)cols);
      else
        hipLaunchKernelGGL((rmsnorm_bwd_kernel<kDT, NT, 32, false>), dim3(grid), dim3(NT), 0,
                           (hipStream_t)stream, (T*)dx, dw_partial, (const T*)dy,
                           nullptr, (const T*)x, (const T*)w, invrms, rows, (int)cols);
    }
  });
  DK_CHECK_LAUNCH();
  return 0;
}

// tmp must hold dk_reduce_tmp_slices(grid) * cols floats
extern "C" int dk_reduce_tmp_slices(int grid) { return (grid + 31) / 32; }

extern "C" int dk_reduce_partials(float* out, float* tmp, const float* partial, int grid,
                                  int64_t cols, dkStream stream) {
  const int nslices = dk_reduce_tmp_slices(grid);
  int cblocks = dk_stream_grid(cols, 256);
  hipLaunchKernelGGL(reduce_partials_stage1, dim3(cblocks, nslices), dim3(256), 0,
                     (hipStream_t)stream, tmp, partial, grid, nslices, cols);
  DK_CHECK_LAUNCH();
  hipLaunchKernelGGL(reduce_partials_stage2, dim3(cblocks), dim3(256), 0,
                     (hipStream_t)stream, out, tmp, nslices, cols);
  DK_CHECK_LAUNCH();
  return 0;
}

// ====================== RoPE ======================
// transformers half-split convention (apply_rotary_pos_emb):
//   out[.., i]      = x1*cos - x2*sin
//   out[.., i+D/2]  = x2*cos + x1*sin        (i < D/2)
// backward is the transposed rotation (sin -> -sin).
// Vectorized over 4 consecutive pairs (8 B loads of each half).

template <int DT, int BWD>
__global__ void rope_kernel(typename DTraits<DT>::T* __restrict__ out,
                            const typename DTraits<DT>::T* __restrict__ x,
                            const float* __restrict__ costab,
                            const float* __restrict__ sintab,
                            int64_t n_rows, int S, int D) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  const int hd = D / 2;
  const int nq = hd / 4;  // quads per row (D/2 divisible by 4 for D in {32,64})
  const int64_t total = n_rows * nq;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / nq;
    const int i0 = (int)(idx % nq) * 4;
    const int pos = (int)(row % S);
    const T* xr = x + row * D;
    T* orow = out + row * D;
    const float4 cv = *(const float4*)(costab + (int64_t)pos * hd + i0);
    const float4 sv = *(const float4*)(sintab + (int64_t)pos * hd + i0);
    shortx4 x1 = *(const shortx4*)(xr + i0);
    shortx4 x2 = *(const shortx4*)(xr + i0 + hd);
    shortx4 o1, o2;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float a = TR::toF(((const T*)&x1)[j]);
      float b = TR::toF(((const T*)&x2)[j]);
      float c = ((const float*)&cv)[j];
      float s = BWD ? -((const float*)&sv)[j] : ((const float*)&sv)[j];
      ((T*)&o1)[j] = TR::fromF(a * c - b * s);
      ((T*)&o2)[j] = TR::fromF(b * c + a * s);
    }
    *(shortx4*)(orow + i0) = o1;
    *(shortx4*)(orow + i0 + hd) = o2;
  }
}

// rope_move: generic-stride RoPE gather/scatter between the fused QKV
// buffer layout (addr = b*sb + h*sh + s*sr + col) and contiguous [B,H,S,D].
// ROT=0 copies without rotation (the V path); BWD applies the transposed
// rotation (gradient path).  Replaces the per-layer transpose+contiguous
// copies AND the separate rope pass in one kernel.
template <int DT, int BWD, int ROT>
__global__ void rope_move_kernel(typename DTraits<DT>::T* __restrict__ out,
                                 const typename DTraits<DT>::T* __restrict__ in,
                                 const float* __restrict__ costab,
                                 const float* __restrict__ sintab,
                                 int64_t total_rows, int H, int S, int D,
                                 int64_t i_sb, int64_t i_sh, int64_t i_sr,
                                 int64_t o_sb, int64_t o_sh, int64_t o_sr) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  const int hd = D / 2;
  const int nq = hd / 8;  // 8-pair pieces per row (b128 loads both halves)
  const int64_t total = total_rows * nq;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / nq;
    const int i0 = (int)(idx % nq) * 8;
    const int sp = (int)(row % S);
    const int hh = (int)((row / S) % H);
    const int64_t b = row / ((int64_t)S * H);
    const T* irow = in + b * i_sb + hh * i_sh + sp * i_sr;
    T* orow = out + b * o_sb + hh * o_sh + sp * o_sr;
    shortx8 x1 = *(const shortx8*)(irow + i0);
    shortx8 x2 = *(const shortx8*)(irow + i0 + hd);
    if (ROT) {
      const float4 cv0 = *(const float4*)(costab + (int64_t)sp * hd + i0);
      const float4 cv1 = *(const float4*)(costab + (int64_t)sp * hd + i0 + 4);
      const float4 sv0 = *(const float4*)(sintab + (int64_t)sp * hd + i0);
      const float4 sv1 = *(const float4*)(sintab + (int64_t)sp * hd + i0 + 4);
      shortx8 o1, o2;
      float af[8], bf[8];
      packed_to_f32<DT, 8>(&x1, af);
      packed_to_f32<DT, 8>(&x2, bf);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float a = af[j];
        float bb = bf[j];
        float c = j < 4 ? ((const float*)&cv0)[j] : ((const float*)&cv1)[j - 4];
        float sn = j < 4 ? ((const float*)&sv0)[j] : ((const float*)&sv1)[j - 4];
        if (BWD) sn = -sn;
        ((T*)&o1)[j] = TR::fromF(a * c - bb * sn);
        ((T*)&o2)[j] = TR::fromF(bb * c + a * sn);
      }
      *(shortx8*)(orow + i0) = o1;
      *(shortx8*)(orow + i0 + hd) = o2;
    } else {
      *(shortx8*)(orow + i0) = x1;
      *(shortx8*)(orow + i0 + hd) = x2;
    }
  }
}

extern "C" int dk_rope_move(void* out, const void* in, const float* costab,
                            const float* sintab, int64_t B, int64_t H, int64_t S,
                            int64_t D, int64_t i_sb, int64_t i_sh, int64_t i_sr,
                            int64_t o_sb, int64_t o_sh, int64_t o_sr, int backward,
                            int rotate, int dtype, dkStream stream) {
  if ((D / 2) % 8 != 0 || dtype == 0) return (int)hipErrorInvalidValue;
  const int64_t total_rows = B * H * S;
  int grid = dk_stream_grid(total_rows * (D / 2) / 8, 256);
  DK_DISPATCH_DT(dtype, {
    if constexpr (kDT != 0) {
      using T = typename DTraits<kDT>::T;
      if (rotate && backward)
        hipLaunchKernelGGL((rope_move_kernel<kDT, 1, 1>), dim3(grid), dim3(256), 0,
                           (hipStream_t)stream, (T*)out, (const T*)in, costab, sintab,
                           total_rows, (int)H, (int)S, (int)D, i_sb, i_sh, i_sr,
                           o_sb, o_sh, o_sr);
      else if (rotate)
        hipLaunchKernelGGL((rope_move_kernel<kDT, 0, 1>), dim3(grid), dim3(256), 0,
                           (hipStream_t)stream, (T*)out, (const T*)in, costab, sintab,
                           total_rows, (int)H, (int)S, (int)D, i_sb, i_sh, i_sr,
                           o_sb, o_sh, o_sr);
      else
        hipLaunchKernelGGL((rope_move_kernel<kDT, 0, 0>), dim3(grid), dim3(256), 0,
                           (hipStream_t)stream, (T*)out, (const T*)in, costab, sintab,
                           total_rows, (int)H, (int)S, (int)D, i_sb, i_sh, i_sr,
                           o_sb, o_sh, o_sr);
    }
  });
  DK_CHECK_LAUNCH();
  return 0;
}

// f32 variant uses scalar pairs (rare path, CPU-parity testing only)
template <int BWD>
__global__ void rope_kernel_f32(float* __restrict__ out, const float* __restrict__ x,
                                const float* __restrict__ costab,
                                const float* __restrict__ sintab,
                                int64_t n_rows, int S, int D) {
  const int hd = D / 2;
  const int64_t total = n_rows * hd;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / hd;
    const int i = (int)(idx % hd);
    const int pos = (int)(row % S);
    float a = x[row * D + i], b = x[row * D + i + hd];
    float c = costab[(int64_t)pos * hd + i];
    float s = BWD ? -sintab[(int64_t)pos * hd + i] : sintab[(int64_t)pos * hd + i];
    out[row * D + i] = a * c - b * s;
    out[row * D + i + hd] = b * c + a * s;
  }
}

extern "C" int dk_rope(void* out, const void* x, const float* costab, const float* sintab,
                       int64_t n_rows, int64_t S, int64_t D, int backward, int dtype,
                       dkStream stream) {
  if ((D / 2) % 4 != 0 && dtype != 0) return (int)hipErrorInvalidValue;
  int grid = dk_stream_grid(n_rows * (D / 2) / (dtype == 0 ? 1 : 4), 256);
  if (dtype == 0) {
    if (backward)
      hipLaunchKernelGGL((rope_kernel_f32<1>), dim3(grid), dim3(256), 0, (hipStream_t)stream,
                         (float*)out, (const float*)x, costab, sintab, n_rows, (int)S, (int)D);
    else
      hipLaunchKernelGGL((rope_kernel_f32<0>), dim3(grid), dim3(256), 0, (hipStream_t)stream,
                         (float*)out, (const float*)x, costab, sintab, n_rows, (int)S, (int)D);
    DK_CHECK_LAUNCH();
    return 0;
  }
  DK_DISPATCH_DT(dtype, {
    if constexpr (kDT != 0) {
      using T = typename DTraits<kDT>::T;
      if (backward)
        hipLaunchKernelGGL((rope_kernel<kDT, 1>), dim3(grid), dim3(256), 0, (hipStream_t)stream,
                           (T*)out, (const T*)x, costab, sintab, n_rows, (int)S, (int)D);
      else
        hipLaunchKernelGGL((rope_kernel<kDT, 0>), dim3(grid), dim3(256), 0, (hipStream_t)stream,
                           (T*)out, (const T*)x, costab, sintab, n_rows, (int)S, (int)D);
    }
  });
  DK_CHECK_LAUNCH();
  return 0;
}

// ====================== SwiGLU ======================
// y = silu(gate) * up;  silu(x) = x * sigmoid(x)  (fp32 internal).

template <int DT>
__global__ void swiglu_fwd_kernel(typename DTraits<DT>::T* __restrict__ y,
                                  const typename DTraits<DT>::T* __restrict__ gate,
                                  const typename DTraits<DT>::T* __restrict__ up,
                                  int64_t nvec) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  constexpr int W = VecIO<T>::W;
  using V = typename VecIO<T>::V;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    V gv = *(const V*)(gate + i * W);
    V uv = *(const V*)(up + i * W);
    V yv;
#pragma unroll
    for (int j = 0; j < W; ++j) {
      float g = TR::toF(((const T*)&gv)[j]);
      float u = TR::toF(((const T*)&uv)[j]);
      float sig = 1.f / (1.f + __expf(-g));
      ((T*)&yv)[j] = TR::fromF(g * sig * u);
    }
    *(V*)(y + i * W) = yv;
  }
}

template <int DT>
__global__ void swiglu_bwd_kernel(typename DTraits<DT>::T* __restrict__ dgate,
                                  typename DTraits<DT>::T* __restrict__ dup,
                                  const typename DTraits<DT>::T* __restrict__ dy,
                                  const typename DTraits<DT>::T* __restrict__ gate,
                                  const typename DTraits<DT>::T* __restrict__ up,
                                  int64_t nvec) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  constexpr int W = VecIO<T>::W;
  using V = typename VecIO<T>::V;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    V gv = *(const V*)(gate + i * W);
    V uv = *(const V*)(up + i * W);
    V dyv = *(const V*)(dy + i * W);
    V dgv, duv;
#pragma unroll
    for (int j = 0; j < W; ++j) {
      float g = TR::toF(((const T*)&gv)[j]);
      float u = TR::toF(((const T*)&uv)[j]);
      float d = TR::toF(((const T*)&dyv)[j]);
      float sig = 1.f / (1.f + __expf(-g));
      float silu = g * sig;
      float dsilu = sig * (1.f + g * (1.f - sig));
      ((T*)&dgv)[j] = TR::fromF(d * u * dsilu);
      ((T*)&duv)[j] = TR::fromF(d * silu);
    }
    *(V*)(dgate + i * W) = dgv;
    *(V*)(dup + i * W) = duv;
  }
}

// fused-layout variant: gate/up live interleaved per row in one [R, 2*I]
// tensor (the batched gate-up GEMM's output); avoids the split+contiguous
// copies entirely.  y[r,c] = silu(gu[r,c]) * gu[r,I+c].
template <int DT>
__global__ void swiglu2_fwd_kernel(typename DTraits<DT>::T* __restrict__ y,
                                   const typename DTraits<DT>::T* __restrict__ gu,
                                   int64_t rows, int64_t nvec_per_row, int64_t I) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  constexpr int W = VecIO<T>::W;
  using V = typename VecIO<T>::V;
  const int64_t total = rows * nvec_per_row;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = idx / nvec_per_row;
    const int64_t c = (idx % nvec_per_row) * W;
    V gv = *(const V*)(gu + r * 2 * I + c);
    V uv = *(const V*)(gu + r * 2 * I + I + c);
    V yv;
#pragma unroll
    for (int j = 0; j < W; ++j) {
      float g = TR::toF(((const T*)&gv)[j]);
      float u = TR::toF(((const T*)&uv)[j]);
      float sig = 1.f / (1.f + __expf(-g));
      ((T*)&yv)[j] = TR::fromF(g * sig * u);
    }
    *(V*)(y + r * I + c) = yv;
  }
}

template <int DT>
__global__ void swiglu2_bwd_kernel(typename DTraits<DT>::T* __restrict__ dgu,
                                   const typename DTraits<DT>::T* __restrict__ dy,
                                   const typename DTraits<DT>::T* __restrict__ gu,
                                   int64_t rows, int64_t nvec_per_row, int64_t I) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  constexpr int W = VecIO<T>::W;
  using V = typename VecIO<T>::V;
  const int64_t total = rows * nvec_per_row;
  for (int64_t idx = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t r = idx / nvec_per_row;
    const int64_t c = (idx % nvec_per_row) * W;
    V gv = *(const V*)(gu + r * 2 * I + c);
    V uv = *(const V*)(gu + r * 2 * I + I + c);
    V dyv = *(const V*)(dy + r * I + c);
    V dgv, duv;
#pragma unroll
    for (int j = 0; j < W; ++j) {
      float g = TR::toF(((const T*)&gv)[j]);
      float u = TR::toF(((const T*)&uv)[j]);
      float d = TR::toF(((const T*)&dyv)[j]);
      float sig = 1.f / (1.f + __expf(-g));
      float silu = g * sig;
      float dsilu = sig * (1.f + g * (1.f - sig));
      ((T*)&dgv)[j] = TR::fromF(d * u * dsilu);
      ((T*)&duv)[j] = TR::fromF(d * silu);
    }
    *(V*)(dgu + r * 2 * I + c) = dgv;
    *(V*)(dgu + r * 2 * I + I + c) = duv;
  }
}

extern "C" int dk_swiglu2_fwd(void* y, const void* gu, int64_t rows, int64_t I,
                              int dtype, dkStream stream) {
  DK_DISPATCH_DT(dtype, {
    using T = typename DTraits<kDT>::T;
    constexpr int W = VecIO<T>::W;
    if (I % W) return (int)hipErrorInvalidValue;
    int grid = dk_stream_grid(rows * (I / W), 256);
    hipLaunchKernelGGL((swiglu2_fwd_kernel<kDT>), dim3(grid), dim3(256), 0,
                       (hipStream_t)stream, (T*)y, (const T*)gu, rows, I / W, I);
  });
  DK_CHECK_LAUNCH();
  return 0;
}

extern "C" int dk_swiglu2_bwd(void* dgu, const void* dy, const void* gu, int64_t rows,
                              int64_t I, int dtype, dkStream stream) {
  DK_DISPATCH_DT(dtype, {
    using T = typename DTraits<kDT>::T;
    constexpr int W = VecIO<T>::W;
    if (I % W) return (int)hipErrorInvalidValue;
    int grid = dk_stream_grid(rows * (I / W), 256);
    hipLaunchKernelGGL((swiglu2_bwd_kernel<kDT>), dim3(grid), dim3(256), 0,
                       (hipStream_t)stream, (T*)dgu, (const T*)dy, (const T*)gu, rows, I / W, I);
  });
  DK_CHECK_LAUNCH();
  return 0;
}

extern "C" int dk_swiglu_fwd(void* y, const void* gate, const void* up, int64_t n,
                             int dtype, dkStream stream) {
  DK_DISPATCH_DT(dtype, {
    using T = typename DTraits<kDT>::T;
    constexpr int W = VecIO<T>::W;
    if (n % W) return (int)hipErrorInvalidValue;
    int grid = dk_stream_grid(n / W, 256);
    hipLaunchKernelGGL((swiglu_fwd_kernel<kDT>), dim3(grid), dim3(256), 0,
                       (hipStream_t)stream, (T*)y, (const T*)gate, (const T*)up, n / W);
  });
  DK_CHECK_LAUNCH();
  return 0;
}

extern "C" int dk_swiglu_bwd(void* dgate, void* dup, const void* dy, const void* gate,
                             const void* up, int64_t n, int dtype, dkStream stream) {
  DK_DISPATCH_DT(dtype, {
    using T = typename DTraits<kDT>::T;
    constexpr int W = VecIO<T>::W;
    if (n % W) return (int)hipErrorInvalidValue;
    int grid = dk_stream_grid(n / W, 256);
    hipLaunchKernelGGL((swiglu_bwd_kernel<kDT>), dim3(grid), dim3(256), 0,
                       (hipStream_t)stream, (T*)dgate, (T*)dup, (const T*)dy,
                       (const T*)gate, (const T*)up, n / W);
  });
  DK_CHECK_LAUNCH();
  return 0;
}

// ====================== Fused AdamW ======================
// torch.optim.AdamW single-tensor op order (decoupled weight decay):
//   p *= 1 - lr*wd
//   m  = m + (1-b1)*(g - m)            (lerp_)
//   v  = b2*v + (1-b2)*g*g
//   p -= (lr/bc1) * m / (sqrt(v)/sqrt(bc2) + eps)
// Flat fp32 state; ~36 B/elem of HBM traffic per step (4 reads + 3 writes).

__global__ void fused_adamw_kernel(float* __restrict__ p, const float* __restrict__ g,
                                   float* __restrict__ m, float* __restrict__ v,
                                   int64_t nvec, float lr, float b1, float b2,
                                   float eps, float wd, float bc1, float bc2_sqrt) {
  const float step_size = lr / bc1;
  const float decay = 1.f - lr * wd;
  const float c1 = 1.f - b1, c2 = 1.f - b2;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    floatx4 pv = *(floatx4*)(p + i * 4);
    floatx4 gv = *(const floatx4*)(g + i * 4);
    floatx4 mv = *(floatx4*)(m + i * 4);
    floatx4 vv = *(floatx4*)(v + i * 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float pj = pv[j] * decay;
      float gj = gv[j];
      float mj = mv[j] + c1 * (gj - mv[j]);
      float vj = b2 * vv[j] + c2 * gj * gj;
      float denom = sqrtf(vj) / bc2_sqrt + eps;
      pv[j] = pj - step_size * mj / denom;
      mv[j] = mj;
      vv[j] = vj;
    }
    *(floatx4*)(p + i * 4) = pv;
    *(floatx4*)(m + i * 4) = mv;
    *(floatx4*)(v + i * 4) = vv;
  }
}

__global__ void fused_adamw_tail_kernel(float* p, const float* g, float* m, float* v,
                                        int64_t start, int64_t n, float lr, float b1,
                                        float b2, float eps, float wd, float bc1,
                                        float bc2_sqrt) {
  const float step_size = lr / bc1;
  const float decay = 1.f - lr * wd;
  int64_t i = start + threadIdx.x;
  if (i < n) {
    float gj = g[i];
    float mj = m[i] + (1.f - b1) * (gj - m[i]);
    float vj = b2 * v[i] + (1.f - b2) * gj * gj;
    p[i] = p[i] * decay - step_size * mj / (sqrtf(vj) / bc2_sqrt + eps);
    m[i] = mj;
    v[i] = vj;
  }
}

extern "C" int dk_fused_adamw(float* p, const float* g, float* m, float* v, int64_t n,
                              float lr, float beta1, float beta2, float eps,
                              float weight_decay, int step, dkStream stream) {
  const float bc1 = 1.f - powf(beta1, (float)step);
  const float bc2_sqrt = sqrtf(1.f - powf(beta2, (float)step));
  int64_t nvec = n / 4;
  int grid = dk_stream_grid(nvec, 256);
  hipLaunchKernelGGL(fused_adamw_kernel, dim3(grid), dim3(256), 0, (hipStream_t)stream,
                     p, g, m, v, nvec, lr, beta1, beta2, eps, weight_decay, bc1, bc2_sqrt);
  DK_CHECK_LAUNCH();
  if (n % 4) {
    hipLaunchKernelGGL(fused_adamw_tail_kernel, dim3(1), dim3(64), 0, (hipStream_t)stream,
                       p, g, m, v, nvec * 4, n, lr, beta1, beta2, eps, weight_decay,
                       bc1, bc2_sqrt);
    DK_CHECK_LAUNCH();
  }
  return 0;
}

// ====================== grad clip (global L2) ======================
// Deterministic two-pass: fixed block->slice assignment, fixed-order reduce.

__global__ void grad_sq_partials_kernel(float* __restrict__ partials,
                                        const float* __restrict__ g, int64_t n) {
  __shared__ float sred[256 / DK_WAVE];
  const int64_t nvec = n / 4;
  float ss = 0.f;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    floatx4 gv = *(const floatx4*)(g + i * 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) ss += gv[j] * gv[j];
  }
  // tail handled by block 0 thread 0..(n%4)
  if (blockIdx.x == 0 && threadIdx.x < (n % 4)) {
    float t = g[nvec * 4 + threadIdx.x];
    ss += t * t;
  }
  ss = block_reduce_sum<256>(ss, sred);
  if (threadIdx.x == 0) partials[blockIdx.x] = ss;
}

__global__ void clip_finalize_kernel(float* __restrict__ out2,
                                     const float* __restrict__ partials, int grid,
                                     float max_norm) {
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int i = 0; i < grid; ++i) s += partials[i];
    float tn = sqrtf(s);
    float coef = max_norm / (tn + 1e-6f);
    out2[0] = tn;
    out2[1] = coef < 1.f ? coef : 1.f;
  }
}

__global__ void clip_scale_kernel(float* __restrict__ g, const float* __restrict__ out2,
                                  int64_t n) {
  const float c = out2[1];
  const int64_t nvec = n / 4;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    floatx4 gv = *(floatx4*)(g + i * 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) gv[j] *= c;
    *(floatx4*)(g + i * 4) = gv;
  }
  if (blockIdx.x == 0 && threadIdx.x < (n % 4)) g[nvec * 4 + threadIdx.x] *= c;
}

extern "C" int dk_gradsq_grid(int64_t n) {
  int g = dk_stream_grid(n / 4, 256);
  return g;
}

extern "C" int dk_grad_sq_partials(float* partials, const float* g, int64_t n, int grid,
                                   dkStream stream) {
  hipLaunchKernelGGL(grad_sq_partials_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, partials, g, n);
  DK_CHECK_LAUNCH();
  return 0;
}

extern "C" int dk_clip_apply(float* g, float* total_norm_out, const float* partials,
                             int grid, int64_t n, float max_norm, dkStream stream) {
  hipLaunchKernelGGL(clip_finalize_kernel, dim3(1), dim3(64), 0, (hipStream_t)stream,
                     total_norm_out, partials, grid, max_norm);
  DK_CHECK_LAUNCH();
  int sg = dk_stream_grid(n / 4, 256);
  hipLaunchKernelGGL(clip_scale_kernel, dim3(sg), dim3(256), 0, (hipStream_t)stream,
                     g, total_norm_out, n);
  DK_CHECK_LAUNCH();
  return 0;
}

// ====================== outer step ======================

__global__ void pseudo_grad_kernel(float* __restrict__ g, const float* __restrict__ to,
                                   const float* __restrict__ tl, int64_t n) {
  const int64_t nvec = n / 4;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    floatx4 a = *(const floatx4*)(to + i * 4);
    floatx4 b = *(const floatx4*)(tl + i * 4);
    floatx4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = a[j] - b[j];
    *(floatx4*)(g + i * 4) = o;
  }
  if (blockIdx.x == 0 && threadIdx.x < (n % 4)) {
    int64_t i = nvec * 4 + threadIdx.x;
    g[i] = to[i] - tl[i];
  }
}

// torch SGD nesterov: buf = first ? g : mu*buf + g; d = g + mu*buf; p -= lr*d
__global__ void outer_nesterov_kernel(float* __restrict__ to, float* __restrict__ tl,
                                      float* __restrict__ buf, const float* __restrict__ g,
                                      int64_t n, float lr, float mu, int first) {
  const int64_t nvec = n / 4;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    floatx4 gv = *(const floatx4*)(g + i * 4);
    floatx4 bv = first ? gv : *(floatx4*)(buf + i * 4);
    floatx4 tv = *(floatx4*)(to + i * 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float b = first ? gv[j] : fmaf(mu, bv[j], gv[j]);
      float d = fmaf(mu, b, gv[j]);
      bv[j] = b;
      tv[j] = tv[j] - lr * d;
    }
    *(floatx4*)(buf + i * 4) = bv;
    *(floatx4*)(to + i * 4) = tv;
    *(floatx4*)(tl + i * 4) = tv;
  }
  if (blockIdx.x == 0 && threadIdx.x < (n % 4)) {
    int64_t i = nvec * 4 + threadIdx.x;
    float b = first ? g[i] : fmaf(mu, buf[i], g[i]);
    float d = fmaf(mu, b, g[i]);
    buf[i] = b;
    to[i] -= lr * d;
    tl[i] = to[i];
  }
}

extern "C" int dk_pseudo_grad(float* g_out, const float* theta_outer,
                              const float* theta_local, int64_t n, dkStream stream) {
  int grid = dk_stream_grid(n / 4, 256);
  hipLaunchKernelGGL(pseudo_grad_kernel, dim3(grid), dim3(256), 0, (hipStream_t)stream,
                     g_out, theta_outer, theta_local, n);
  DK_CHECK_LAUNCH();
  return 0;
}

extern "C" int dk_outer_nesterov(float* theta_outer, float* theta_local,
                                 float* momentum_buf, const float* g_avg, int64_t n,
                                 float lr, float momentum, int first_step,
                                 dkStream stream) {
  int grid = dk_stream_grid(n / 4, 256);
  hipLaunchKernelGGL(outer_nesterov_kernel, dim3(grid), dim3(256), 0, (hipStream_t)stream,
                     theta_outer, theta_local, momentum_buf, g_avg, n, lr, momentum,
                     first_step);
  DK_CHECK_LAUNCH();
  return 0;
}

// ====================== casts ======================

template <int DDT, int SDT>
__global__ void cast_kernel(typename DTraits<DDT>::T* __restrict__ dst,
                            const typename DTraits<SDT>::T* __restrict__ src, int64_t n) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] = DTraits<DDT>::fromF(DTraits<SDT>::toF(src[i]));
}

extern "C" int dk_cast(void* dst, const void* src, int64_t n, int dst_dtype,
                       int src_dtype, dkStream stream) {
  int grid = dk_stream_grid(n, 256);
  DK_DISPATCH_DT(dst_dtype, {
    constexpr int kD = kDT;
    DK_DISPATCH_DT(src_dtype, {
      hipLaunchKernelGGL((cast_kernel<kD, kDT>), dim3(grid), dim3(256), 0,
                         (hipStream_t)stream, (typename DTraits<kD>::T*)dst,
                         (const typename DTraits<kDT>::T*)src, n);
    });
  });
  DK_CHECK_LAUNCH();
  return 0;
}

// fused gradient accumulation: fp32 master grad += low-precision dW, one
// kernel instead of the cast + autograd-add pair (and no intermediate fp32
// dW tensor round-trip).  Vectorized 8-wide on both sides.
template <int SDT>
__global__ void accum_kernel(float* __restrict__ dst,
                             const typename DTraits<SDT>::T* __restrict__ src,
                             int64_t n) {
  using TR = DTraits<SDT>;
  using T = typename TR::T;
  typedef __attribute__((ext_vector_type(8))) short vec8;
  typedef __attribute__((ext_vector_type(4))) float fvec4;
  const int64_t nv = n / 8;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    vec8 sv = *(const vec8*)(src + i * 8);
    fvec4 d0 = *(const fvec4*)(dst + i * 8);
    fvec4 d1 = *(const fvec4*)(dst + i * 8 + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      d0[j] += TR::toF(((const T*)&sv)[j]);
      d1[j] += TR::toF(((const T*)&sv)[4 + j]);
    }
    *(fvec4*)(dst + i * 8) = d0;
    *(fvec4*)(dst + i * 8 + 4) = d1;
  }
  for (int64_t i = nv * 8 + blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] += TR::toF(src[i]);
}

extern "C" int dk_accum(float* dst, const void* src, int64_t n, int src_dtype,
                        dkStream stream) {
  int grid = dk_stream_grid(n / 8 + 1, 256);
  DK_DISPATCH_DT(src_dtype, {
    if constexpr (kDT != 0) {
      hipLaunchKernelGGL((accum_kernel<kDT>), dim3(grid), dim3(256), 0,
                         (hipStream_t)stream, dst,
                         (const typename DTraits<kDT>::T*)src, n);
    } else {
      return (int)hipErrorInvalidValue;
    }
  });
  DK_CHECK_LAUNCH();
  return 0;
}

// ---- split-K dW chunk reduction ----
// dst[i] += sum_b src[b * chunk_stride + i]: deterministic (fixed summation
// order over the split-K partial slabs of the weight-gradient GEMM), fp32
// end to end, fused with the master-grad accumulate — replaces the
// reference's per-weight cast + autograd-add (grads of F.linear inside
// transformers Llama, called at train_fsdp.py:383).
__global__ void accum_chunks_kernel(float* __restrict__ dst,
                                    const float* __restrict__ src,
                                    int64_t n, int nchunk, int64_t chunk_stride) {
  typedef __attribute__((ext_vector_type(4))) float fvec4;
  const int64_t nv = n / 4;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    fvec4 d = *(const fvec4*)(dst + i * 4);
    for (int b = 0; b < nchunk; ++b) {
      fvec4 s = *(const fvec4*)(src + b * chunk_stride + i * 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) d[j] += s[j];
    }
    *(fvec4*)(dst + i * 4) = d;
  }
  for (int64_t i = nv * 4 + blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float d = dst[i];
    for (int b = 0; b < nchunk; ++b) d += src[b * chunk_stride + i];
    dst[i] = d;
  }
}

extern "C" int dk_accum_chunks(float* dst, const float* src, int64_t n, int nchunk,
                               int64_t chunk_stride, dkStream stream) {
  int grid = dk_stream_grid(n / 4 + 1, 256);
  hipLaunchKernelGGL(accum_chunks_kernel, dim3(grid), dim3(256), 0,
                     (hipStream_t)stream, dst, src, n, nchunk, chunk_stride);
  DK_CHECK_LAUNCH();
  return 0;
}

extern "C" const char* dk_version(void) { return "diloco_kernels gfx950 0.1.0"; }
