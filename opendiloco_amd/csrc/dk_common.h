// dk_common.h — shared device helpers for the MI355X (gfx950) DiLoCo kernels.
// CDNA4: wavefront = 64 lanes; block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#define DK_WAVE 64

typedef __attribute__((ext_vector_type(4))) float floatx4;
typedef __attribute__((ext_vector_type(8))) short shortx8;
typedef __attribute__((ext_vector_type(4))) short shortx4;

// ---------- scalar dtype conversions ----------
__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)u) << 16;
  return c.f;
}

__device__ __forceinline__ unsigned short f32_to_bf16(float f) {
  // native cast lowers to ONE v_cvt_pk_bf16_f32 (RTNE on gfx950; adjacent
  // casts pair into a single pk instruction).  The previous manual
  // round-to-nearest-even (+0x7fff+lsb shift chain, ~6 VALU) cost the
  // issue-bound attention kernels 15-25% — measured as the f16-vs-bf16
  // gap in profiles/round2: identical kernels, v_cvt_f16 vs manual bf16.
  __bf16 h = (__bf16)f;
  unsigned short u;
  __builtin_memcpy(&u, &h, 2);
  return u;
}

// ---------- dtype traits: DT 0=f32, 1=f16, 2=bf16 ----------
template <int DT> struct DTraits;

template <> struct DTraits<0> {
  using T = float;
  static __device__ __forceinline__ float toF(T x) { return x; }
  static __device__ __forceinline__ T fromF(float x) { return x; }
};
template <> struct DTraits<1> {
  using T = _Float16;
  static __device__ __forceinline__ float toF(T x) { return (float)x; }
  static __device__ __forceinline__ T fromF(float x) { return (_Float16)x; }
};
template <> struct DTraits<2> {
  using T = unsigned short;  // bf16 bit pattern
  static __device__ __forceinline__ float toF(T x) { return bf16_to_f32(x); }
  static __device__ __forceinline__ T fromF(float x) { return f32_to_bf16(x); }
};

// Packed bf16-pair upconvert: one dword holding 2 bf16 becomes 2 floats in
// TWO VALU ops (lshl + and) — the scalar path costs an extra 16-bit extract
// per element (measured 10-20% on the HBM-bound read-heavy kernels vs the
// f16 variant, whose v_cvt_f32_f16 takes a 16-bit source select).
__device__ __forceinline__ void bf16x2_to_f32(unsigned int u, float& lo, float& hi) {
  union { unsigned int i; float f; } a, b;
  a.i = u << 16;
  b.i = u & 0xffff0000u;
  lo = a.f;
  hi = b.f;
}

// convert N packed 16-bit elements (N even, dword-aligned vector) to float[N]
template <int DT, int N>
__device__ __forceinline__ void packed_to_f32(const void* v, float* out) {
  static_assert(N % 2 == 0, "pairwise");
  if constexpr (DT == 2) {
#pragma unroll
    for (int p = 0; p < N / 2; ++p)
      bf16x2_to_f32(((const unsigned int*)v)[p], out[2 * p], out[2 * p + 1]);
  } else if constexpr (DT == 1) {
#pragma unroll
    for (int j = 0; j < N; ++j)
      out[j] = (float)((const _Float16*)v)[j];
  } else {
#pragma unroll
    for (int j = 0; j < N; ++j)
      out[j] = ((const float*)v)[j];
  }
}

// vector-of-8 load/store for 16-bit types (16 B — the coalescing sweet spot,
// cdna_hip_programming.md G13), vector-of-4 for f32.
template <typename T> struct VecIO;
template <> struct VecIO<float> {
  static constexpr int W = 4;
  using V = floatx4;
};
template <> struct VecIO<_Float16> {
  static constexpr int W = 8;
  using V = shortx8;
};
template <> struct VecIO<unsigned short> {
  static constexpr int W = 8;
  using V = shortx8;
};

// ---------- wave + block reductions ----------
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, DK_WAVE);
  return x;
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, DK_WAVE));
  return x;
}

// block reduce over NT threads (NT multiple of 64, <= 1024); smem: float[NT/64]
template <int NT>
__device__ __forceinline__ float block_reduce_sum(float x, float* smem) {
  constexpr int NW = NT / DK_WAVE;
  const int wid = threadIdx.x / DK_WAVE;
  x = wave_reduce_sum(x);
  if ((threadIdx.x & (DK_WAVE - 1)) == 0) smem[wid] = x;
  __syncthreads();
  float r = 0.f;
#pragma unroll
  for (int i = 0; i < NW; ++i) r += smem[i];
  __syncthreads();
  return r;
}

// two independent sums through ONE barrier (amortizes the per-row block
// sync of row-per-block kernels when two rows are processed per iteration);
// smem: float[2 * NT/64]
template <int NT>
__device__ __forceinline__ void block_reduce_sum2(float& a, float& b, float* smem) {
  constexpr int NW = NT / DK_WAVE;
  const int wid = threadIdx.x / DK_WAVE;
  a = wave_reduce_sum(a);
  b = wave_reduce_sum(b);
  if ((threadIdx.x & (DK_WAVE - 1)) == 0) { smem[wid] = a; smem[NW + wid] = b; }
  __syncthreads();
  float ra = 0.f, rb = 0.f;
#pragma unroll
  for (int i = 0; i < NW; ++i) { ra += smem[i]; rb += smem[NW + i]; }
  __syncthreads();
  a = ra;
  b = rb;
}

template <int NT>
__device__ __forceinline__ float block_reduce_max(float x, float* smem) {
  constexpr int NW = NT / DK_WAVE;
  const int wid = threadIdx.x / DK_WAVE;
  x = wave_reduce_max(x);
  if ((threadIdx.x & (DK_WAVE - 1)) == 0) smem[wid] = x;
  __syncthreads();
  float r = -INFINITY;
#pragma unroll
  for (int i = 0; i < NW; ++i) r = fmaxf(r, smem[i]);
  __syncthreads();
  return r;
}

#define DK_CHECK_LAUNCH() \
  do { hipError_t e_ = hipGetLastError(); if (e_ != hipSuccess) return (int)e_; } while (0)

// dispatch over dtype (0=f32,1=f16,2=bf16)
#define DK_DISPATCH_DT(dtype, ...)                    \
  switch (dtype) {                                    \
    case 0: { constexpr int kDT = 0; __VA_ARGS__; break; } \
    case 1: { constexpr int kDT = 1; __VA_ARGS__; break; } \
    case 2: { constexpr int kDT = 2; __VA_ARGS__; break; } \
    default: return (int)hipErrorInvalidValue;        \
  }

// grid sizing for memory-bound grid-stride kernels (G11: cap ~2048 blocks)
static inline int dk_stream_grid(int64_t work_items, int block) {
  int64_t g = (work_items + block - 1) / block;
  if (g > 2048) g = 2048;
  if (g < 1) g = 1;
  return (int)g;
}
