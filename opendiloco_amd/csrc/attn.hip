// attn.hip — causal flash attention fwd/bwd for gfx950 (CDNA4 MFMA).
//
// Replaces torch SDPA inside the reference's Llama forward
// (attn_implementation "sdpa", reference train_fsdp.py:107; called via
// model(**batch) at train_fsdp.py:378 / train_diloco_torch.py:313) and its
// autograd backward (train_fsdp.py:383).
//
// Design (MI355X-first, correctness-first structure for round 1):
//  - mfma_f32_16x16x32_{bf16,f16} tiles; 4 waves (256 thr) per workgroup.
//  - forward: workgroup owns 64 q rows of one (b, h); waves own 16 q rows
//    each; K/V staged in LDS by all 256 threads (K row-major [32][D+8] for
//    direct B-fragment ds_read_b128, V transposed [D][32+8] at staging time
//    so PV B-fragments are contiguous reads); online softmax in fp32 with
//    running (m, l) per row, wave-shuffle row reductions (no serial lanes);
//    P goes through a small per-wave LDS tile to re-shape C-layout ->
//    A-layout.  lse = m + log(l) saved for backward.
//  - backward: split into a dK/dV kernel (grid over kv tiles; P^T recomputed
//    from lse) and a dQ kernel (grid over q tiles) — no atomics anywhere, so
//    gradients are bit-deterministic run to run (the reference's tests
//    compare loss traces, test_train.py:82).  GQA handled by writing dK/dV
//    per q-head; the caller sums the group (ratio == 1 writes directly).
//
// MFMA fragment maps used (verified on hardware by dk_probe_mfma_16x16x32,
// tests/test_gpu_ops.py::test_mfma_probe):
//   A[16][32]: lane l holds row = l&15,  k = 8*(l>>4) + j   (j = 0..7)
//   B[32][16]: lane l holds col = l&15,  k = 8*(l>>4) + j
//   C[16][16]: lane l holds col = l&15,  row = (l>>4)*4 + r (r = 0..3)

#include "dk_common.h"
#include "../../include/diloco_kernels.h"

#include <math.h>
#include <stdlib.h>

typedef __attribute__((ext_vector_type(8))) _Float16 halfx8;

template <int DT> struct MFMA16;
template <> struct MFMA16<2> {
  using frag = shortx8;
  static __device__ __forceinline__ floatx4 mma(frag a, frag b, floatx4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  }
};
template <> struct MFMA16<1> {
  using frag = halfx8;
  static __device__ __forceinline__ floatx4 mma(frag a, frag b, floatx4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
  }
};

// gfx950 hardware transpose-read: within each 16-lane group, the lanes
// supply the 8-byte chunks of a [4 rows][16 cols] bf16/f16 tile and lane L
// receives column L&15: out[j] = tile[row j][col L&15].  Lane addresses are
// independent, so the tile rows may live at ANY stride — a 16x16x32 MFMA
// B-fragment can be read straight from a ROW-MAJOR LDS image (two reads:
// k = 8g..8g+3 and 8g+4..8g+7, g = lane>>4), eliminating the separate
// transposed staging image and its 8 scalar LDS writes per thread.
// Chunk->lane semantics hardware-verified (tools: tr16_probe).
template <int DT> struct TrRead;
template <> struct TrRead<2> {
  typedef __attribute__((ext_vector_type(4))) __bf16 v4;
  static __device__ __forceinline__ void rd(const unsigned short* p, void* out) {
    v4 r = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (__attribute__((address_space(3))) v4*)(uintptr_t)(const void*)p);
    __builtin_memcpy(out, &r, 8);
  }
};
template <> struct TrRead<1> {
  typedef __attribute__((ext_vector_type(4))) __fp16 v4;
  static __device__ __forceinline__ void rd(const _Float16* p, void* out) {
    v4 r = __builtin_amdgcn_ds_read_tr16_b64_v4f16(
        (__attribute__((address_space(3))) v4*)(uintptr_t)(const void*)p);
    __builtin_memcpy(out, &r, 8);
  }
};
// B-fragment (16x16x32 map: lane l -> col l&15, k = 8*(l>>4)+j) from a
// row-major image img[row][ds]: rows row0.., cols col0..col0+15.
template <int DT>
__device__ __forceinline__ typename MFMA16<DT>::frag trread_bfrag(
    const typename DTraits<DT>::T* img, int row0, int col0, int ds, int lane) {
  const int rb = row0 + 8 * (lane >> 4) + ((lane & 15) >> 2);
  const int cb = col0 + 4 * (lane & 3);
  typename MFMA16<DT>::frag f;
  TrRead<DT>::rd(img + rb * ds + cb, &f);
  TrRead<DT>::rd(img + (rb + 4) * ds + cb, (char*)&f + 8);
  return f;
}

// 4-wide 16-bit vector store (one ds_write_b64) for the packed P/dS images
template <int DT> struct Pack4;
template <> struct Pack4<2> { using V = shortx4; };
template <> struct Pack4<1> { typedef __attribute__((ext_vector_type(4))) _Float16 V; };

#define NEG_BIG (-1e30f)

// row-group shuffle reduce: combine over the 16 lanes that share l>>4
__device__ __forceinline__ float grp16_max(float x) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) x = fmaxf(x, __shfl_xor(x, off, DK_WAVE));
  return x;
}
__device__ __forceinline__ float grp16_sum(float x) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) x += __shfl_xor(x, off, DK_WAVE);
  return x;
}

// ======================= forward =======================
// v2 structure (guide T14/T3/T5): KT=64 kv tiles, double-buffered LDS, async
// register staging (next tile's global loads issued before compute, written
// to the other LDS buffer after it — HBM latency hides under MFMA), ONE
// barrier per tile, both K and V staged ROW-MAJOR (PV fragments are
// transposed at read time by ds_read_b64_tr_b16), s_setprio(1) around the
// MFMA clusters.
// LDS: K[2][KT][DS] | V[2][KT][DS] | P[4][16][KS]

template <int DT, int D>
__global__ __launch_bounds__(256) void attn_fwd_kernel(
    typename DTraits<DT>::T* __restrict__ o, float* __restrict__ lse,
    const typename DTraits<DT>::T* __restrict__ q,
    const typename DTraits<DT>::T* __restrict__ k,
    const typename DTraits<DT>::T* __restrict__ v,
    int B, int Hq, int Hkv, int S, float scale,
    int64_t o_sb, int64_t o_sh, int64_t o_sr,
    int64_t v_sb, int64_t v_sh, int64_t v_sr) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  using MF = MFMA16<DT>;
  using frag = typename MF::frag;
  constexpr int KT = 64;           // kv tile
  constexpr int KS = KT + 8;       // padded LDS stride (keys dim), u16 units
  constexpr int DS = D + 8;        // padded LDS stride (channel dim)
  constexpr int NKC = D / 32;      // mfma K-chunks over channels
  constexpr int NDN = D / 16;      // output channel tiles
  constexpr int NNT = KT / 16;     // 16-key sub-tiles per kv tile
  constexpr int LPT = (KT * D) / 8 / 256;  // b128 loads per thread per tile

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  T* K_lds = (T*)smem_raw;                       // [2][KT][DS]
  T* V_lds = K_lds + 2 * KT * DS;                // [2][KT][DS] row-major
  T* P_lds = V_lds + 2 * KT * DS;                // [4][16][KS]

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo = lane & 15;        // col index (key / d-channel)
  const int hi = lane >> 4;        // k-chunk group & row group

  const int nQT = (S + 63) / 64;
  int bid = blockIdx.x;
  const int qt = bid % nQT;
  const int h = (bid / nQT) % Hq;
  const int b = bid / (nQT * Hq);
  const int hkv = h / (Hq / Hkv);

  const int q0 = qt * 64 + wave * 16;            // this wave's first q row
  const int64_t qoff = (((int64_t)b * Hq + h) * S) * D;
  const int64_t kvoff = (((int64_t)b * Hkv + hkv) * S) * D;
  const int64_t voff = v_sb ? ((int64_t)b * v_sb + (int64_t)hkv * v_sh) : kvoff;
  const int64_t v_rs = v_sb ? v_sr : (int64_t)D;

  // Q A-fragments for this wave's 16 rows (clamped on the tail tile)
  frag q_frag[NKC];
  {
    const int qrow = q0 + lo;
    const int qr_c = qrow < S ? qrow : S - 1;
#pragma unroll
    for (int kc = 0; kc < NKC; ++kc)
      q_frag[kc] = *(const frag*)(q + qoff + (int64_t)qr_c * D + kc * 32 + hi * 8);
  }

  float m_run[4], l_run[4];
  floatx4 o_acc[NDN];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = NEG_BIG; l_run[r] = 0.f; }
#pragma unroll
  for (int dn = 0; dn < NDN; ++dn) o_acc[dn] = (floatx4)(0.f);

  const int kv_end = min(S, qt * 64 + 64);       // causal upper bound
  const int n_kt = (kv_end + KT - 1) / KT;

  // staging registers (async split): this thread's pieces of the next tile
  frag kreg[LPT], vreg[LPT];
  const int st_row[2] = {(int)threadIdx.x / (D / 8), (int)(threadIdx.x + 256) / (D / 8)};
  const int st_c8[2] = {((int)threadIdx.x % (D / 8)) * 8, ((int)(threadIdx.x + 256) % (D / 8)) * 8};

  auto load_tile = [&](int kt) {
#pragma unroll
    for (int i = 0; i < LPT; ++i) {
      const int krow = kt * KT + st_row[i];
      const int kr_c = krow < S ? krow : S - 1;
      kreg[i] = *(const frag*)(k + kvoff + (int64_t)kr_c * D + st_c8[i]);
      vreg[i] = *(const frag*)(v + voff + (int64_t)kr_c * v_rs + st_c8[i]);
    }
  };
  auto write_tile = [&](int buf) {
    T* Kb = K_lds + buf * KT * DS;
    T* Vb = V_lds + buf * KT * DS;
#pragma unroll
    for (int i = 0; i < LPT; ++i) {
      *(frag*)(Kb + st_row[i] * DS + st_c8[i]) = kreg[i];
      *(frag*)(Vb + st_row[i] * DS + st_c8[i]) = vreg[i];
    }
  };

  load_tile(0);
  write_tile(0);
  __syncthreads();

  for (int kt = 0; kt < n_kt; ++kt) {
    const int kbase = kt * KT;
    const int cur = kt & 1;
    T* Kb = K_lds + cur * KT * DS;
    T* Vb = V_lds + cur * KT * DS;
    if (kt + 1 < n_kt) load_tile(kt + 1);  // async: in flight during compute

    // ---- S tile = Q K^T (16 q x KT keys) ----
    floatx4 sc[NNT];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int nt = 0; nt < NNT; ++nt) {
      sc[nt] = (floatx4)(0.f);
#pragma unroll
      for (int kc = 0; kc < NKC; ++kc) {
        frag bk = *(const frag*)(Kb + (nt * 16 + lo) * DS + kc * 32 + hi * 8);
        sc[nt] = MF::mma(q_frag[kc], bk, sc[nt]);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- mask + online softmax over KT keys ----
    // tiles strictly below the diagonal and fully in-bounds need no
    // per-element mask (wave-uniform branch; saves 2 cmp+sel per element)
    const bool full_tile = (kbase + KT <= q0) && (kbase + KT <= S);
    float p[NNT][4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0 + hi * 4 + r;
      float rm = NEG_BIG;
      float sv[NNT];
      if (full_tile) {
#pragma unroll
        for (int nt = 0; nt < NNT; ++nt) {
          sv[nt] = sc[nt][r] * scale;
          rm = fmaxf(rm, sv[nt]);
        }
      } else {
#pragma unroll
        for (int nt = 0; nt < NNT; ++nt) {
          const int kk = kbase + nt * 16 + lo;
          sv[nt] = (kk > qrow || kk >= S) ? NEG_BIG : sc[nt][r] * scale;
          rm = fmaxf(rm, sv[nt]);
        }
      }
      rm = grp16_max(rm);
      const float m_new = fmaxf(m_run[r], rm);
      const float alpha = __expf(m_run[r] - m_new);
      float psum = 0.f;
#pragma unroll
      for (int nt = 0; nt < NNT; ++nt) {
        p[nt][r] = sv[nt] <= NEG_BIG ? 0.f : __expf(sv[nt] - m_new);
        psum += p[nt][r];
      }
      l_run[r] = l_run[r] * alpha + grp16_sum(psum);
      m_run[r] = m_new;
#pragma unroll
      for (int dn = 0; dn < NDN; ++dn) o_acc[dn][r] *= alpha;
    }

    // ---- P through per-wave LDS: C-layout -> A-layout ----
    T* Pw = P_lds + wave * 16 * KS;
#pragma unroll
    for (int nt = 0; nt < NNT; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r) Pw[(hi * 4 + r) * KS + nt * 16 + lo] = TR::fromF(p[nt][r]);
    // same-wave LDS dependency: compiler inserts lgkmcnt waits
    frag pa[KT / 32];
#pragma unroll
    for (int c = 0; c < KT / 32; ++c)
      pa[c] = *(const frag*)(Pw + lo * KS + c * 32 + hi * 8);

    // ---- O += P V ----
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int dn = 0; dn < NDN; ++dn)
#pragma unroll
      for (int c = 0; c < KT / 32; ++c) {
        frag bv = trread_bfrag<DT>(Vb, c * 32, dn * 16, DS, lane);
        o_acc[dn] = MF::mma(pa[c], bv, o_acc[dn]);
      }
    __builtin_amdgcn_s_setprio(0);

    if (kt + 1 < n_kt) write_tile((kt + 1) & 1);  // T14: write late, after compute
    __syncthreads();
  }

  // ---- epilogue ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + hi * 4 + r;
    if (qrow >= S) continue;
    const float inv_l = l_run[r] > 0.f ? 1.f / l_run[r] : 0.f;
    const int64_t obase = (int64_t)b * o_sb + (int64_t)h * o_sh + (int64_t)qrow * o_sr;
#pragma unroll
    for (int dn = 0; dn < NDN; ++dn)
      o[obase + dn * 16 + lo] = TR::fromF(o_acc[dn][r] * inv_l);
    if (lo == 0) lse[((int64_t)b * Hq + h) * S + qrow] = m_run[r] + __logf(l_run[r]);
  }
}

// ======================= forward v3: swapped QK^T, in-register softmax =======================
// 32x32x16 MFMA tiles.  S^T = mfma(K, Q) puts 16 of a q-row's 32 key-scores
// in ONE lane (q = lane&31; the other 16 in the partner lane l^32), so the
// softmax row reduction is an in-lane tree + ONE shfl_xor(32) — no 16-lane
// shuffle chains, no P LDS round trip: P is packed to bf16 in-register
// (cvt_pk pairs) and redistributed with permlane32_swap into the PV B-frag
// (layouts + swap semantics hardware-verified by dk_probe_mfma_32x32x16 and
// dk_probe_permlane32).  PV computes O^T = mfma(V^T, P) so the accumulator
// stays q-lane-local (alpha rescale in-lane).  Wave owns 32 q rows; LDS is
// only the double-buffered K/V^T tiles (same async staging as v2).
typedef __attribute__((ext_vector_type(16))) float floatx16;
typedef __attribute__((ext_vector_type(4))) int intx4;

template <int DT> struct MFMA32;
template <> struct MFMA32<2> {
  using frag = shortx8;
  static __device__ __forceinline__ floatx16 mma(frag a, frag b, floatx16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  }
};
template <> struct MFMA32<1> {
  using frag = halfx8;
  static __device__ __forceinline__ floatx16 mma(frag a, frag b, floatx16 c) {
    return __builtin_amdgcn_mfma_f32_32x32x16_f16(a, b, c, 0, 0, 0);
  }
};

// A-fragment for 32x32x16 MFMA (map: lane l -> row l&31, k = 8*(l>>5)+j)
// transposed out of a ROW-MAJOR image img[k][ds] via ds_read_b64_tr_b16:
// out[j] = img[row0 + 8*(l>>5) + j][col0 + (l&31)].  The four 16-lane tr
// groups split as (k-half = l>>5) x (col-half = (l>>4)&1).
template <int DT>
__device__ __forceinline__ typename MFMA32<DT>::frag trread_afrag32(
    const typename DTraits<DT>::T* img, int row0, int col0, int ds, int lane) {
  const int lam = lane & 15;
  const int rb = row0 + 8 * (lane >> 5) + (lam >> 2);
  const int cb = col0 + 16 * ((lane >> 4) & 1) + 4 * (lam & 3);
  typename MFMA32<DT>::frag f;
  TrRead<DT>::rd(img + rb * ds + cb, &f);
  TrRead<DT>::rd(img + (rb + 4) * ds + cb, (char*)&f + 8);
  return f;
}

template <int DT, int D>
__global__ __launch_bounds__(256, 4) void attn_fwd_v3_kernel(
    typename DTraits<DT>::T* __restrict__ o, float* __restrict__ lse,
    const typename DTraits<DT>::T* __restrict__ q,
    const typename DTraits<DT>::T* __restrict__ k,
    const typename DTraits<DT>::T* __restrict__ v,
    int B, int Hq, int Hkv, int S, float scale,
    int64_t o_sb, int64_t o_sh, int64_t o_sr,
    int64_t v_sb, int64_t v_sh, int64_t v_sr) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  using MF = MFMA32<DT>;
  using frag = typename MF::frag;
  constexpr int KT = 64;            // kv tile (2 x 32-key subtiles)
  constexpr int KS = KT + 8;        // P image stride (u16)
  constexpr int DS = D + 8;         // K image stride
  constexpr int NKC = D / 16;       // 16-channel contraction chunks for QK^T
  constexpr int NMT = D / 32;       // 32-row d tiles for O^T
  constexpr int LPT = (KT * D) / 8 / 256;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  T* K_lds = (T*)smem_raw;                       // [2][KT][DS]
  T* V_lds = K_lds + 2 * KT * DS;                // [2][KT][DS] row-major

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo32 = lane & 31;       // q row within the wave
  const int hi5 = lane >> 5;

  const int nQT = (S + 127) / 128;
  int bid = blockIdx.x;
  const int qt = bid % nQT;
  const int h = (bid / nQT) % Hq;
  const int b = bid / (nQT * Hq);
  const int hkv = h / (Hq / Hkv);

  const int q0 = qt * 128 + wave * 32;           // wave's first q row
  const int qrow = q0 + lo32;                    // this lane's q row
  const int64_t qoff = (((int64_t)b * Hq + h) * S) * D;
  const int64_t kvoff = (((int64_t)b * Hkv + hkv) * S) * D;
  const int64_t voff = v_sb ? ((int64_t)b * v_sb + (int64_t)hkv * v_sh) : kvoff;
  const int64_t v_rs = v_sb ? v_sr : (int64_t)D;

  // persistent Q B-fragments: slot j of chunk kc = Q[qrow][kc*16 + hi5*8 + j]
  frag q_frag[NKC];
  {
    const int qr_c = qrow < S ? qrow : S - 1;
#pragma unroll
    for (int kc = 0; kc < NKC; ++kc)
      q_frag[kc] = *(const frag*)(q + qoff + (int64_t)qr_c * D + kc * 16 + hi5 * 8);
  }

  float m_run = NEG_BIG, l_run = 0.f;
  floatx16 oacc[NMT];
#pragma unroll
  for (int mt = 0; mt < NMT; ++mt) oacc[mt] = (floatx16)(0.f);

  const int kv_end = min(S, qt * 128 + 128);
  const int n_kt = (kv_end + KT - 1) / KT;

  // async double-buffered staging (same structure as v2)
  shortx8 kreg[LPT], vreg[LPT];
  const int st_row[2] = {(int)threadIdx.x / (D / 8), (int)(threadIdx.x + 256) / (D / 8)};
  const int st_c8[2] = {((int)threadIdx.x % (D / 8)) * 8, ((int)(threadIdx.x + 256) % (D / 8)) * 8};

  auto load_tile = [&](int kt) {
#pragma unroll
    for (int i = 0; i < LPT; ++i) {
      const int krow = kt * KT + st_row[i];
      const int kr_c = krow < S ? krow : S - 1;
      kreg[i] = *(const shortx8*)(k + kvoff + (int64_t)kr_c * D + st_c8[i]);
      vreg[i] = *(const shortx8*)(v + voff + (int64_t)kr_c * v_rs + st_c8[i]);
    }
  };
  auto write_tile = [&](int buf) {
    T* Kb = K_lds + buf * KT * DS;
    T* Vb = V_lds + buf * KT * DS;
#pragma unroll
    for (int i = 0; i < LPT; ++i) {
      *(shortx8*)(Kb + st_row[i] * DS + st_c8[i]) = kreg[i];
      *(shortx8*)(Vb + st_row[i] * DS + st_c8[i]) = vreg[i];
    }
  };

  load_tile(0);
  write_tile(0);
  __syncthreads();

  for (int kt = 0; kt < n_kt; ++kt) {
    const int kbase = kt * KT;
    const int cur = kt & 1;
    T* Kb = K_lds + cur * KT * DS;
    T* Vb = V_lds + cur * KT * DS;
    if (kt + 1 < n_kt) load_tile(kt + 1);

#pragma unroll
    for (int st = 0; st < 2; ++st) {            // two 32-key subtiles
      const int sbase = kbase + st * 32;
      // wave-uniform diagonal skip: keys all above this wave's q rows would
      // be fully masked (rm = NEG_BIG -> alpha 1, psum 0: state unchanged).
      if (sbase >= q0 + 32) continue;
      // ---- S^T = K Q^T ----
      floatx16 sc = (floatx16)(0.f);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kc = 0; kc < NKC; ++kc) {
        frag ka = *(const frag*)(Kb + (st * 32 + lo32) * DS + kc * 16 + hi5 * 8);
        sc = MF::mma(ka, q_frag[kc], sc);
      }
      __builtin_amdgcn_s_setprio(0);

      // ---- in-lane online softmax for q row `qrow` (p values live in the
      // sc accumulator registers: no extra p[] array) ----
      const bool full_tile = (sbase + 32 <= q0) && (sbase + 32 <= S);
      float rm = NEG_BIG;
      if (full_tile) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          sc[r] *= scale;
          rm = fmaxf(rm, sc[r]);
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int key = sbase + (r & 3) + 8 * (r >> 2) + 4 * hi5;
          sc[r] = (key > qrow || key >= S) ? NEG_BIG : sc[r] * scale;
          rm = fmaxf(rm, sc[r]);
        }
      }
      rm = fmaxf(rm, __shfl_xor(rm, 32, DK_WAVE));
      const float m_new = fmaxf(m_run, rm);
      const float alpha = __expf(m_run - m_new);
      float psum = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        sc[r] = sc[r] <= NEG_BIG ? 0.f : __expf(sc[r] - m_new);
        psum += sc[r];
      }
      psum += __shfl_xor(psum, 32, DK_WAVE);
      l_run = l_run * alpha + psum;
      m_run = m_new;
#pragma unroll
      for (int mt = 0; mt < NMT; ++mt)
#pragma unroll
        for (int r = 0; r < 16; ++r) oacc[mt][r] *= alpha;

      // ---- pack P into PV B-fragments via cvt_pk + permlane32_swap ----
      // chunk c covers keys sbase + c*16 .. +15; per chunk the four packed
      // words are w0=(k0,k1) w1=(k2,k3) w2=(k8,k9) w3=(k10,k11) in the low
      // half (+4 in the high half); swap(w0,w2) -> frag dwords {d0, d2},
      // swap(w1,w3) -> {d1, d3}  (hardware-verified regrouping).
      auto bits = [](T t) -> unsigned {
        unsigned short u;
        __builtin_memcpy(&u, &t, 2);
        return (unsigned)u;
      };
      frag pfrag[2];
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        const int base = c * 8;
        unsigned w0 = bits(TR::fromF(sc[base + 0])) | (bits(TR::fromF(sc[base + 1])) << 16);
        unsigned w1 = bits(TR::fromF(sc[base + 2])) | (bits(TR::fromF(sc[base + 3])) << 16);
        unsigned w2 = bits(TR::fromF(sc[base + 4])) | (bits(TR::fromF(sc[base + 5])) << 16);
        unsigned w3 = bits(TR::fromF(sc[base + 6])) | (bits(TR::fromF(sc[base + 7])) << 16);
        auto s0 = __builtin_amdgcn_permlane32_swap((int)w0, (int)w2, false, false);
        auto s1 = __builtin_amdgcn_permlane32_swap((int)w1, (int)w3, false, false);
        intx4 pw;
        pw[0] = s0[0];
        pw[1] = s1[0];
        pw[2] = s0[1];
        pw[3] = s1[1];
        pfrag[c] = *(frag*)&pw;
      }

      // ---- O^T += V^T P  (A = V^T tr_read from row-major V, B = P) ----
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mt = 0; mt < NMT; ++mt)
#pragma unroll
        for (int c = 0; c < 2; ++c) {
          frag va = trread_afrag32<DT>(Vb, st * 32 + c * 16, mt * 32, DS,
                                       (int)(threadIdx.x & 63));
          oacc[mt] = MF::mma(va, pfrag[c], oacc[mt]);
        }
      __builtin_amdgcn_s_setprio(0);
    }

    if (kt + 1 < n_kt) write_tile((kt + 1) & 1);
    __syncthreads();
  }

  // ---- epilogue: lane q = lo32 owns its whole row of O^T ----
  if (qrow < S) {
    const float inv_l = l_run > 0.f ? 1.f / l_run : 0.f;
    const int64_t obase = (int64_t)b * o_sb + (int64_t)h * o_sh + (int64_t)qrow * o_sr;
#pragma unroll
    for (int mt = 0; mt < NMT; ++mt)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = mt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi5;
        o[obase + d] = TR::fromF(oacc[mt][r] * inv_l);
      }
    if (hi5 == 0) lse[((int64_t)b * Hq + h) * S + qrow] = m_run + __logf(l_run);
  }
}

// ======================= bwd preprocess: delta = rowsum(do*o) =======================
// each wave covers WR=4 rows per iteration: 16 lanes per row, shortx4 loads
// (8 B/lane), 4-lane... 16-lane-group shuffle reduce per row
template <int DT>
__global__ void attn_bwd_pre_kernel(float* __restrict__ delta,
                                    const typename DTraits<DT>::T* __restrict__ do_,
                                    const typename DTraits<DT>::T* __restrict__ o,
                                    int64_t rows, int H, int S, int D,
                                    int64_t g_sb, int64_t g_sh, int64_t g_sr) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo = lane & 15, hi = lane >> 4;   // lo: D-quad index, hi: row-in-group
  const int64_t w0 = (int64_t)blockIdx.x * 4 + wave;
  const int64_t stride = (int64_t)gridDim.x * 4;
  const int quads = D / 4;  // shortx4 chunks per row (D in {32, 64})
  const int64_t groups = (rows + 3) / 4;
  for (int64_t g = w0; g < groups; g += stride) {  // 4 rows per wave-iteration
    const int64_t r = g * 4 + hi;
    const int64_t rc = r < rows ? r : rows - 1;
    const int sp = (int)(rc % S);
    const int hh = (int)((rc / S) % H);
    const int64_t b = rc / ((int64_t)S * H);
    const int64_t base = b * g_sb + hh * g_sh + sp * g_sr;
    float s = 0.f;
    for (int q4 = lo; q4 < quads; q4 += 16) {
      shortx4 dv = *(const shortx4*)(do_ + base + q4 * 4);
      shortx4 ov = *(const shortx4*)(o + base + q4 * 4);
      float df[4], of[4];
      packed_to_f32<DT, 4>(&dv, df);
      packed_to_f32<DT, 4>(&ov, of);
#pragma unroll
      for (int j = 0; j < 4; ++j) s += df[j] * of[j];
    }
#pragma unroll
    for (int off = 1; off < 16; off <<= 1) s += __shfl_xor(s, off, DK_WAVE);
    if (lo == 0 && r < rows) delta[r] = s;
  }
}

// ======================= bwd dK/dV =======================
// grid over (b, hq, kv-tile of 128 keys); 8 waves x 16 keys each.  Loop q
// tiles of 32 with double-buffered async staging (v2 structure); dV/dK
// B-fragments tr_read straight from the row-major Q/dO images.
// LDS: Q[2][32][D+8] | dO[2][32][D+8] | lse[2][32] f32 | delta[2][32] f32 |
//      P_T[8][16][32+8] | dS[8][16][32+8]   (dV/dK B-frags come straight
//      from the row-major Q/dO images via ds_read_b64_tr_b16)
template <int DT, int D>
__global__ __launch_bounds__(512) void attn_bwd_dkdv_kernel(
    typename DTraits<DT>::T* __restrict__ dk_out,
    typename DTraits<DT>::T* __restrict__ dv_out,
    const typename DTraits<DT>::T* __restrict__ do_,
    const typename DTraits<DT>::T* __restrict__ q,
    const typename DTraits<DT>::T* __restrict__ k,
    const typename DTraits<DT>::T* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    int B, int Hq, int Hkv, int S, float scale,
    int64_t g_sb, int64_t g_sh, int64_t g_sr,
    int64_t v_sb, int64_t v_sh, int64_t v_sr,
    int64_t dv_sb, int64_t dv_sh, int64_t dv_sr) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  using MF = MFMA16<DT>;
  using frag = typename MF::frag;
  constexpr int QT = 32;                     // compute half-tile (q rows)
  constexpr int QTT = 64;                    // staged tile: 2 halves / barrier
  constexpr int QS = QT + 8;
  constexpr int DS = D + 8;
  constexpr int NKC = D / 32;
  constexpr int NDN = D / 16;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  T* Q_lds = (T*)smem_raw;                   // [2][QTT][DS]
  T* dO_lds = Q_lds + 2 * QTT * DS;          // [2][QTT][DS]
  T* PT_lds = dO_lds + 2 * QTT * DS;         // [8][16][QS]  (P^T tiles)
  T* DS_lds = PT_lds + 8 * 16 * QS;          // [8][16][QS]  (dS^T tiles; separate
                                             //  buffer: avoids an LDS WAR hazard
                                             //  between the P^T A-frag read and
                                             //  the dS^T writes in one iteration)
  float* lse_lds = (float*)(DS_lds + 8 * 16 * QS);  // [2][QTT]
  float* dl_lds = lse_lds + 2 * QTT;                // [2][QTT]

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo = lane & 15;
  const int hi = lane >> 4;

  const int nKT = (S + 127) / 128;           // 8 waves x 16 keys per WG
  int bid = blockIdx.x;
  const int kt = bid % nKT;
  const int h = (bid / nKT) % Hq;
  const int b = bid / (nKT * Hq);
  const int hkv = h / (Hq / Hkv);

  const int k0 = kt * 128 + wave * 16;       // this wave's first key
  const int64_t qoff = (((int64_t)b * Hq + h) * S) * D;
  const int64_t kvoff = (((int64_t)b * Hkv + hkv) * S) * D;
  const int64_t lseoff = ((int64_t)b * Hq + h) * S;
  const int64_t voff = v_sb ? ((int64_t)b * v_sb + (int64_t)hkv * v_sh) : kvoff;
  const int64_t v_rs = v_sb ? v_sr : (int64_t)D;

  // K,V A-fragments for this wave's 16 keys
  frag k_frag[NKC], v_frag[NKC];
  {
    const int krow = k0 + lo;
    const int kr_c = krow < S ? krow : S - 1;
#pragma unroll
    for (int kc = 0; kc < NKC; ++kc) {
      k_frag[kc] = *(const frag*)(k + kvoff + (int64_t)kr_c * D + kc * 32 + hi * 8);
      v_frag[kc] = *(const frag*)(v + voff + (int64_t)kr_c * v_rs + kc * 32 + hi * 8);
    }
  }

  floatx4 dv_acc[NDN], dk_acc[NDN];
#pragma unroll
  for (int dn = 0; dn < NDN; ++dn) { dv_acc[dn] = (floatx4)(0.f); dk_acc[dn] = (floatx4)(0.f); }

  const int qstart = (kt * 128) / QTT;       // first staged tile with these keys
  const int nQT2 = (S + QTT - 1) / QTT;

  // async staging state (QTT*D/8 = 512 loads: one piece per thread)
  const int st_t = (int)threadIdx.x;
  const bool st_on = st_t < (QTT * D) / 8;
  const int st_row = st_t / (D / 8);
  const int st_c8 = (st_t % (D / 8)) * 8;
  frag qreg, dreg;
  float lse_reg = 0.f, dl_reg = 0.f;

  const int64_t gbase = (int64_t)b * g_sb + (int64_t)h * g_sh;
  auto load_qtile = [&](int qt) {
    const int qrow = qt * QTT + st_row;
    const int qr_c = qrow < S ? qrow : S - 1;
    if (st_on) {
      qreg = *(const frag*)(q + qoff + (int64_t)qr_c * D + st_c8);
      dreg = *(const frag*)(do_ + gbase + (int64_t)qr_c * g_sr + st_c8);
    }
    if (st_t < QTT) {
      const int rr = qt * QTT + st_t;
      const int rr_c = rr < S ? rr : S - 1;
      lse_reg = lse[lseoff + rr_c];
      dl_reg = delta[lseoff + rr_c];
    }
  };
  auto write_qtile = [&](int buf) {
    if (st_on) {
      *(frag*)(Q_lds + buf * QTT * DS + st_row * DS + st_c8) = qreg;
      *(frag*)(dO_lds + buf * QTT * DS + st_row * DS + st_c8) = dreg;
    }
    if (st_t < QTT) {
      lse_lds[buf * QTT + st_t] = lse_reg;
      dl_lds[buf * QTT + st_t] = dl_reg;
    }
  };

  load_qtile(qstart);
  write_qtile(0);
  __syncthreads();

  for (int qt = qstart; qt < nQT2; ++qt) {
    const int cur = (qt - qstart) & 1;
    T* Q64 = Q_lds + cur * QTT * DS;
    T* dO64 = dO_lds + cur * QTT * DS;
    const float* lse64 = lse_lds + cur * QTT;
    const float* dl64 = dl_lds + cur * QTT;
    if (qt + 1 < nQT2) load_qtile(qt + 1);

    // two 32-q compute halves per staged 64-row tile: one barrier per 64 q
#pragma unroll
    for (int hf = 0; hf < 2; ++hf) {
    const int qbase = qt * QTT + hf * QT;
    T* Qb = Q64 + hf * QT * DS;
    T* dOb = dO64 + hf * QT * DS;
    const float* lse_b = lse64 + hf * QT;
    const float* dl_b = dl64 + hf * QT;
    // 8-wave WGs skew the diagonal: a wave whose 16 keys all sit above this
    // q half (k0 > every qcol) computes an all-masked (zero) tile — skip.
    if (qbase + QT > k0) {
    // ---- S = Q K^T computed TRANSPOSED-C: mma(Q_as_A, K_as_B) ----
    // The 16x16x32 A and B fragments share one lane map (row/col = l&15,
    // k = 8*(l>>4)+j), so swapping the operands flips the C orientation
    // for free: C row = q (hi*4+r), col = key (lo).  Each lane then holds
    // 4 CONSECUTIVE q values for one key, which pack into the P^T/dS^T
    // images ([16 key][QS] row-major, exactly the round-1 layout) as ONE
    // ds_write_b64 instead of 16 scalar ds_write_b16 per image; the
    // A-fragment reads stay the round-1 contiguous b128 reads, and the
    // per-q lse/delta lookups become one float4 read per 16-q block.
    float pt[2][4], dst[2][4];
    __builtin_amdgcn_s_setprio(1);
    const int krow = k0 + lo;
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      floatx4 st = (floatx4)(0.f);
      floatx4 dpt = (floatx4)(0.f);
#pragma unroll
      for (int kc = 0; kc < NKC; ++kc) {
        frag qa = *(const frag*)(Qb + (nt * 16 + lo) * DS + kc * 32 + hi * 8);
        st = MF::mma(qa, k_frag[kc], st);
        frag doa = *(const frag*)(dOb + (nt * 16 + lo) * DS + kc * 32 + hi * 8);
        dpt = MF::mma(doa, v_frag[kc], dpt);
      }
      const floatx4 lse4 = *(const floatx4*)(lse_b + nt * 16 + hi * 4);
      const floatx4 dl4 = *(const floatx4*)(dl_b + nt * 16 + hi * 4);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = qbase + nt * 16 + hi * 4 + r;
        float pv = 0.f;
        if (krow <= qrow && krow < S && qrow < S)
          pv = __expf(st[r] * scale - lse4[r]);
        pt[nt][r] = pv;
        dst[nt][r] = pv * (dpt[r] - dl4[r]) * scale;
      }
    }

    // ---- dV += P^T dO  (A = P^T via packed per-wave LDS image; B = dO_T) ----
    T* Pw = PT_lds + wave * 16 * QS;
    using P4 = typename Pack4<DT>::V;
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      P4 pk;
#pragma unroll
      for (int r = 0; r < 4; ++r) ((T*)&pk)[r] = TR::fromF(pt[nt][r]);
      *(P4*)(Pw + lo * QS + nt * 16 + hi * 4) = pk;
    }
    frag pa = *(const frag*)(Pw + lo * QS + hi * 8);
#pragma unroll
    for (int dn = 0; dn < NDN; ++dn) {
      frag bd = trread_bfrag<DT>(dOb, 0, dn * 16, DS, lane);
      dv_acc[dn] = MF::mma(pa, bd, dv_acc[dn]);
    }

    // ---- dK += dS^T Q  (A = dS^T via the same packed image layout) ----
    T* Dw = DS_lds + wave * 16 * QS;
#pragma unroll
    for (int nt = 0; nt < 2; ++nt) {
      P4 dk4;
#pragma unroll
      for (int r = 0; r < 4; ++r) ((T*)&dk4)[r] = TR::fromF(dst[nt][r]);
      *(P4*)(Dw + lo * QS + nt * 16 + hi * 4) = dk4;
    }
    frag da = *(const frag*)(Dw + lo * QS + hi * 8);
#pragma unroll
    for (int dn = 0; dn < NDN; ++dn) {
      frag bq = trread_bfrag<DT>(Qb, 0, dn * 16, DS, lane);
      dk_acc[dn] = MF::mma(da, bq, dk_acc[dn]);
    }
    __builtin_amdgcn_s_setprio(0);
    }  // end diagonal skip
    }  // end half loop

    if (qt + 1 < nQT2) write_qtile(cur ^ 1);  // T14: write late
    __syncthreads();
  }

  // ---- write dK, dV (per q-head layout [B,Hq,S,D]; caller sums GQA groups) ----
  const int64_t dvoff = dv_sb ? ((int64_t)b * dv_sb + (int64_t)h * dv_sh) : qoff;
  const int64_t dv_rs = dv_sb ? dv_sr : (int64_t)D;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int krow = k0 + hi * 4 + r;
    if (krow >= S) continue;
#pragma unroll
    for (int dn = 0; dn < NDN; ++dn) {
      dk_out[qoff + (int64_t)krow * D + dn * 16 + lo] = TR::fromF(dk_acc[dn][r]);
      dv_out[dvoff + (int64_t)krow * dv_rs + dn * 16 + lo] = TR::fromF(dv_acc[dn][r]);
    }
  }
}

// ======================= bwd dQ =======================
// grid over (b, hq, q-tile of 128); 8 waves x 16 q rows each.  Loop kv
// tiles of 64; dQ B-fragments tr_read from the row-major K image.
// LDS: K[2][64][D+8] | V[2][64][D+8] | dS[8][16][64+8]
template <int DT, int D>
__global__ __launch_bounds__(512) void attn_bwd_dq_kernel(
    typename DTraits<DT>::T* __restrict__ dq_out,
    const typename DTraits<DT>::T* __restrict__ do_,
    const typename DTraits<DT>::T* __restrict__ q,
    const typename DTraits<DT>::T* __restrict__ k,
    const typename DTraits<DT>::T* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    int B, int Hq, int Hkv, int S, float scale,
    int64_t g_sb, int64_t g_sh, int64_t g_sr,
    int64_t v_sb, int64_t v_sh, int64_t v_sr) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  using MF = MFMA16<DT>;
  using frag = typename MF::frag;
  constexpr int KT = 64;                      // kv tile (halves barrier count)
  constexpr int KS = KT + 8;
  constexpr int DS = D + 8;
  constexpr int NKC = D / 32;
  constexpr int NDN = D / 16;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  T* K_lds = (T*)smem_raw;                    // [3][KT][DS] (triple ring)
  T* V_lds = K_lds + 3 * KT * DS;             // [3][KT][DS]
  T* S_lds = V_lds + 3 * KT * DS;             // [8][16][KS]  (dQ B-frags come
                                              //  from row-major K via tr_read)

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo = lane & 15;
  const int hi = lane >> 4;

  const int nQT = (S + 127) / 128;            // 8 waves x 16 q rows per WG
  int bid = blockIdx.x;
  const int qt = bid % nQT;
  const int h = (bid / nQT) % Hq;
  const int b = bid / (nQT * Hq);
  const int hkv = h / (Hq / Hkv);

  const int q0 = qt * 128 + wave * 16;
  const int64_t qoff = (((int64_t)b * Hq + h) * S) * D;
  const int64_t kvoff = (((int64_t)b * Hkv + hkv) * S) * D;
  const int64_t lseoff = ((int64_t)b * Hq + h) * S;
  const int64_t voff = v_sb ? ((int64_t)b * v_sb + (int64_t)hkv * v_sh) : kvoff;
  const int64_t v_rs = v_sb ? v_sr : (int64_t)D;

  frag q_frag[NKC], do_frag[NKC];
  float lse_q, dl_q;
  const int qrow_l = q0 + lo;   // this lane's q row (C col after the operand swap)
  {
    const int qr_c = qrow_l < S ? qrow_l : S - 1;
    const int64_t gbase = (int64_t)b * g_sb + (int64_t)h * g_sh;
#pragma unroll
    for (int kc = 0; kc < NKC; ++kc) {
      q_frag[kc] = *(const frag*)(q + qoff + (int64_t)qr_c * D + kc * 32 + hi * 8);
      do_frag[kc] = *(const frag*)(do_ + gbase + (int64_t)qr_c * g_sr + kc * 32 + hi * 8);
    }
    lse_q = lse[lseoff + qr_c];
    dl_q = delta[lseoff + qr_c];
  }

  floatx4 dq_acc[NDN];
#pragma unroll
  for (int dn = 0; dn < NDN; ++dn) dq_acc[dn] = (floatx4)(0.f);

  const int kv_end = min(S, qt * 128 + 128);
  const int n_kt = (kv_end + KT - 1) / KT;

  // async TRIPLE-buffered staging (prefetch distance 2: the t+2 load is
  // issued at the START of tile t, the t+1 registers land in LDS at the
  // END of tile t — ~2 compute tiles of load slack).  Two register sets
  // (A = even tiles, B = odd) cost +8 VGPRs (120 total, still 4
  // waves/SIMD); 3 LDS buffers fit 2 WGs/CU (147 KB).  MEASURED: a wash
  // vs double-buffering (209 vs 212 TF) — dq's 50% parked
  // (round2_attn_pmc.md) is LDS-read latency inside the mma loops and
  // barrier convoy, not global staging slack.  Kept: equal speed,
  // deeper slack for other shapes.
  const int st_t = (int)threadIdx.x;
  const bool st_on = st_t < (KT * D) / 8;
  const int st_row = st_t / (D / 8);
  const int st_c8 = (st_t % (D / 8)) * 8;
  frag kregA, vregA, kregB, vregB;

  auto load_ktile = [&](int kt, frag& kr, frag& vr) {
    if (st_on) {
      const int krow = kt * KT + st_row;
      const int kr_c = krow < S ? krow : S - 1;
      kr = *(const frag*)(k + kvoff + (int64_t)kr_c * D + st_c8);
      vr = *(const frag*)(v + voff + (int64_t)kr_c * v_rs + st_c8);
    }
  };
  auto write_ktile = [&](int buf, const frag& kr, const frag& vr) {
    if (st_on) {
      *(frag*)(K_lds + buf * KT * DS + st_row * DS + st_c8) = kr;
      *(frag*)(V_lds + buf * KT * DS + st_row * DS + st_c8) = vr;
    }
  };

  load_ktile(0, kregA, vregA);
  write_ktile(0, kregA, vregA);
  if (1 < n_kt) load_ktile(1, kregB, vregB);
  __syncthreads();

  for (int kt = 0; kt < n_kt; ++kt) {
    const int kbase = kt * KT;
    const int cur = kt % 3;
    T* Kb = K_lds + cur * KT * DS;
    T* Vb = V_lds + cur * KT * DS;
    if (kt + 2 < n_kt) {
      if ((kt + 2) & 1) load_ktile(kt + 2, kregB, vregB);
      else load_ktile(kt + 2, kregA, vregA);
    }

    // wave-uniform diagonal skip (8-wave skew): if every key in this kv
    // tile exceeds this wave's last q row, the whole tile is masked to zero.
    if (kbase < q0 + 16) {
    // S computed TRANSPOSED-C (see the dkdv kernel): mma(K_as_A, Q_as_B)
    // puts C row = key (hi*4+r), col = q (lo), so each lane holds 4
    // CONSECUTIVE keys of one q row — the dS^T image ([16 q][KS] row-major,
    // round-1 layout) takes ONE packed ds_write_b64 per 16-key block
    // instead of 16 scalar ds_write_b16, and the A-fragment reads stay the
    // round-1 contiguous b128 reads.
    float ds[4][4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      floatx4 st = (floatx4)(0.f), dpt = (floatx4)(0.f);
#pragma unroll
      for (int kc = 0; kc < NKC; ++kc) {
        frag ka = *(const frag*)(Kb + (nt * 16 + lo) * DS + kc * 32 + hi * 8);
        st = MF::mma(ka, q_frag[kc], st);
        frag va = *(const frag*)(Vb + (nt * 16 + lo) * DS + kc * 32 + hi * 8);
        dpt = MF::mma(va, do_frag[kc], dpt);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kcol = kbase + nt * 16 + hi * 4 + r;
        float pv = 0.f;
        if (kcol <= qrow_l && kcol < S && qrow_l < S)
          pv = __expf(st[r] * scale - lse_q);
        ds[nt][r] = pv * (dpt[r] - dl_q) * scale;
      }
    }

    // dQ += dS K  (A = dS via the packed [16 q][KS] image; B from
    // row-major K via tr_read); KT=64 keys = two 32-deep contraction chunks
    T* Sw = S_lds + wave * 16 * KS;
    using P4 = typename Pack4<DT>::V;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      P4 s4;
#pragma unroll
      for (int r = 0; r < 4; ++r) ((T*)&s4)[r] = TR::fromF(ds[nt][r]);
      *(P4*)(Sw + lo * KS + nt * 16 + hi * 4) = s4;
    }
    frag da[2];
#pragma unroll
    for (int c = 0; c < 2; ++c) da[c] = *(const frag*)(Sw + lo * KS + c * 32 + hi * 8);
#pragma unroll
    for (int dn = 0; dn < NDN; ++dn)
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        frag bk = trread_bfrag<DT>(Kb, c * 32, dn * 16, DS, lane);
        dq_acc[dn] = MF::mma(da[c], bk, dq_acc[dn]);
      }
    __builtin_amdgcn_s_setprio(0);
    }  // end diagonal skip

    if (kt + 1 < n_kt) {  // T14: write late (registers loaded last iteration)
      if ((kt + 1) & 1) write_ktile((kt + 1) % 3, kregB, vregB);
      else write_ktile((kt + 1) % 3, kregA, vregA);
    }
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + hi * 4 + r;
    if (qrow >= S) continue;
#pragma unroll
    for (int dn = 0; dn < NDN; ++dn)
      dq_out[qoff + (int64_t)qrow * D + dn * 16 + lo] = TR::fromF(dq_acc[dn][r]);
  }
}

// ======================= bwd v3: 32x32 MFMA tiles =======================
// Same split (dK/dV over kv tiles, dQ over q tiles) and the same staged
// images as v2, but on mfma_f32_32x32x16 tiles: a wave owns 32 keys (dkdv)
// or 32 q rows (dq) and the workgroup covers 128, halving the MFMA / LDS
// fragment-read instruction count per FLOP.  P^T / dS^T still re-shape
// C->A through per-wave LDS tiles (a half-lane swap cannot transpose them).

template <int DT, int D>
__global__ __launch_bounds__(256) void attn_bwd_dkdv_v3_kernel(
    typename DTraits<DT>::T* __restrict__ dk_out,
    typename DTraits<DT>::T* __restrict__ dv_out,
    const typename DTraits<DT>::T* __restrict__ do_,
    const typename DTraits<DT>::T* __restrict__ q,
    const typename DTraits<DT>::T* __restrict__ k,
    const typename DTraits<DT>::T* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    int B, int Hq, int Hkv, int S, float scale,
    int64_t g_sb, int64_t g_sh, int64_t g_sr) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  using MF = MFMA32<DT>;
  using frag = typename MF::frag;
  constexpr int QT = 32;
  constexpr int QS = QT + 8;
  constexpr int DS = D + 8;
  constexpr int NKC = D / 16;
  constexpr int NMT = D / 32;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  T* Q_lds = (T*)smem_raw;                   // [2][QT][DS]
  T* dO_lds = Q_lds + 2 * QT * DS;           // [2][QT][DS]
  T* PT_lds = dO_lds + 2 * QT * DS;          // [4][32][QS]
  T* DST_lds = PT_lds + 4 * 32 * QS;         // [4][32][QS]
  float* lse_lds = (float*)(DST_lds + 4 * 32 * QS);  // [2][QT]
  float* dl_lds = lse_lds + 2 * QT;                  // [2][QT]

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo32 = lane & 31;
  const int hi5 = lane >> 5;

  const int nKT = (S + 127) / 128;
  int bid = blockIdx.x;
  const int kt = bid % nKT;
  const int h = (bid / nKT) % Hq;
  const int b = bid / (nKT * Hq);
  const int hkv = h / (Hq / Hkv);

  const int k0 = kt * 128 + wave * 32;       // wave's first key
  const int64_t qoff = (((int64_t)b * Hq + h) * S) * D;
  const int64_t kvoff = (((int64_t)b * Hkv + hkv) * S) * D;
  const int64_t lseoff = ((int64_t)b * Hq + h) * S;

  // K,V A-fragment base (A row = key = lane&31); fragments are re-read per
  // q-tile from L2 instead of held in registers — frees 32 VGPRs so the
  // kernel fits 3 waves/SIMD without spills
  const int kr_c0 = (k0 + lo32) < S ? (k0 + lo32) : S - 1;
  const T* kbase_p = k + kvoff + (int64_t)kr_c0 * D + hi5 * 8;
  const T* vbase_p = v + kvoff + (int64_t)kr_c0 * D + hi5 * 8;

  floatx16 dv_acc[NMT], dk_acc[NMT];
#pragma unroll
  for (int mt = 0; mt < NMT; ++mt) { dv_acc[mt] = (floatx16)(0.f); dk_acc[mt] = (floatx16)(0.f); }

  const int qstart = (kt * 128) / QT;
  const int nQT2 = (S + QT - 1) / QT;

  const int st_t = (int)threadIdx.x;
  const bool st_on = st_t < (QT * D) / 8;
  const int st_row = st_t / (D / 8);
  const int st_c8 = (st_t % (D / 8)) * 8;
  shortx8 qreg, dreg;
  float lse_reg = 0.f, dl_reg = 0.f;
  const int64_t gbase = (int64_t)b * g_sb + (int64_t)h * g_sh;

  auto load_qtile = [&](int qt2) {
    const int qrow = qt2 * QT + st_row;
    const int qr_c = qrow < S ? qrow : S - 1;
    if (st_on) {
      qreg = *(const shortx8*)(q + qoff + (int64_t)qr_c * D + st_c8);
      dreg = *(const shortx8*)(do_ + gbase + (int64_t)qr_c * g_sr + st_c8);
    }
    if (st_t < QT) {
      const int rr = qt2 * QT + st_t;
      const int rr_c = rr < S ? rr : S - 1;
      lse_reg = lse[lseoff + rr_c];
      dl_reg = delta[lseoff + rr_c];
    }
  };
  auto write_qtile = [&](int buf) {
    if (st_on) {
      *(shortx8*)(Q_lds + buf * QT * DS + st_row * DS + st_c8) = qreg;
      *(shortx8*)(dO_lds + buf * QT * DS + st_row * DS + st_c8) = dreg;
    }
    if (st_t < QT) {
      lse_lds[buf * QT + st_t] = lse_reg;
      dl_lds[buf * QT + st_t] = dl_reg;
    }
  };

  load_qtile(qstart);
  write_qtile(0);
  __syncthreads();

  for (int qt2 = qstart; qt2 < nQT2; ++qt2) {
    const int qbase = qt2 * QT;
    const int cur = (qt2 - qstart) & 1;
    T* Qb = Q_lds + cur * QT * DS;
    T* dOb = dO_lds + cur * QT * DS;
    const float* lse_b = lse_lds + cur * QT;
    const float* dl_b = dl_lds + cur * QT;
    if (qt2 + 1 < nQT2) load_qtile(qt2 + 1);

    // ---- S^T = K Q^T and dP^T = V dO^T  (C: col = q = lane&31) ----
    floatx16 st = (floatx16)(0.f), dpt = (floatx16)(0.f);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kc = 0; kc < NKC; ++kc) {
      frag ka = *(const frag*)(kbase_p + kc * 16);
      frag bq = *(const frag*)(Qb + lo32 * DS + kc * 16 + hi5 * 8);
      st = MF::mma(ka, bq, st);
      frag va = *(const frag*)(vbase_p + kc * 16);
      frag bd = *(const frag*)(dOb + lo32 * DS + kc * 16 + hi5 * 8);
      dpt = MF::mma(va, bd, dpt);
    }
    __builtin_amdgcn_s_setprio(0);

    const int qcol = qbase + lo32;
    const float lse_q = lse_b[lo32];
    const float dl_q = dl_b[lo32];
    const bool full = (qbase >= k0 + 32) && (qbase + QT <= S);
    float pt[16], dst[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      float pv;
      if (full) {
        pv = __expf(st[r] * scale - lse_q);
      } else {
        const int krow = k0 + (r & 3) + 8 * (r >> 2) + 4 * hi5;
        pv = 0.f;
        if (krow <= qcol && krow < S && qcol < S)
          pv = __expf(st[r] * scale - lse_q);
      }
      pt[r] = pv;
      dst[r] = pv * (dpt[r] - dl_q) * scale;
    }

    // ---- dV += P^T dO (A = P^T via LDS; B = dO_T) ----
    T* Pw = PT_lds + wave * 32 * QS;
#pragma unroll
    for (int r = 0; r < 16; ++r)
      Pw[((r & 3) + 8 * (r >> 2) + 4 * hi5) * QS + lo32] = TR::fromF(pt[r]);
    frag pa[2];
#pragma unroll
    for (int c = 0; c < 2; ++c)
      pa[c] = *(const frag*)(Pw + lo32 * QS + c * 16 + hi5 * 8);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mt = 0; mt < NMT; ++mt)
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        frag bd = trread_afrag32<DT>(dOb, c * 16, mt * 32, DS, lane);
        dv_acc[mt] = MF::mma(pa[c], bd, dv_acc[mt]);
      }
    __builtin_amdgcn_s_setprio(0);

    // ---- dK += dS^T Q (A = dS^T via LDS; B = Q_T) ----
    T* Dw = DST_lds + wave * 32 * QS;
#pragma unroll
    for (int r = 0; r < 16; ++r)
      Dw[((r & 3) + 8 * (r >> 2) + 4 * hi5) * QS + lo32] = TR::fromF(dst[r]);
    frag da[2];
#pragma unroll
    for (int c = 0; c < 2; ++c)
      da[c] = *(const frag*)(Dw + lo32 * QS + c * 16 + hi5 * 8);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mt = 0; mt < NMT; ++mt)
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        frag bq = trread_afrag32<DT>(Qb, c * 16, mt * 32, DS, lane);
        dk_acc[mt] = MF::mma(da[c], bq, dk_acc[mt]);
      }
    __builtin_amdgcn_s_setprio(0);

    if (qt2 + 1 < nQT2) write_qtile(cur ^ 1);
    __syncthreads();
  }

  // ---- write dK, dV (per q-head layout; caller sums GQA groups) ----
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int krow = k0 + (r & 3) + 8 * (r >> 2) + 4 * hi5;
    if (krow >= S) continue;
#pragma unroll
    for (int mt = 0; mt < NMT; ++mt) {
      dk_out[qoff + (int64_t)krow * D + mt * 32 + lo32] = TR::fromF(dk_acc[mt][r]);
      dv_out[qoff + (int64_t)krow * D + mt * 32 + lo32] = TR::fromF(dv_acc[mt][r]);
    }
  }
}

// ======================= bwd split-32: dv-only / dk-only =======================
// The fused kernels carry BOTH accumulator sets (dV and dK) and are
// VGPR-capped at 4 (16x16) / 2 (32x32) waves per SIMD while the matrix
// pipe idles at ~11% (profiles/round2_attn_pmc.md) — issue/latency bound.
// Splitting halves the accumulators: each kernel holds ONE floatx16[D/32]
// set on 32x32x16 tiles (2x the FLOP per instruction of the 16x16 path),
// recomputes S from lse, and occupies 4-5 waves/SIMD.  QK^T is recomputed
// by both kernels (+25% matrix-pipe work) against a ~40% cut in issue
// slots per FLOP.  The mma runs OPERAND-SWAPPED (A/B lane maps coincide)
// so each lane's C holds 4 consecutive q per key: the P^T / dS^T staging
// image takes 4 packed ds_write_b64 instead of 16 scalar ds_write_b16 and
// the lse/delta lookups are float4 reads (same trick as the 16x16 pair).
// Grid over (b, hq, kv-tile of 128); 4 waves x 32 keys; q-tiles of 32,
// double-buffered.  Strided dO/V/dV supported (runs in-model, unlike the
// contiguous-only fused 32x32 port).  WANT_DK=0: dV += P^T dO.
// WANT_DK=1: dK += dS^T Q with dS^T = P^T o (dP^T - delta) * scale.
template <int DT, int D, int WANT_DK>
__global__ __launch_bounds__(256) void attn_bwd_split32_kernel(
    typename DTraits<DT>::T* __restrict__ out,      // dV or dK
    const typename DTraits<DT>::T* __restrict__ do_,
    const typename DTraits<DT>::T* __restrict__ q,
    const typename DTraits<DT>::T* __restrict__ k,
    const typename DTraits<DT>::T* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    int B, int Hq, int Hkv, int S, float scale,
    int64_t g_sb, int64_t g_sh, int64_t g_sr,
    int64_t v_sb, int64_t v_sh, int64_t v_sr,
    int64_t o_sb, int64_t o_sh, int64_t o_sr) {   // out strides (0 = BHSD)
  using TR = DTraits<DT>;
  using T = typename TR::T;
  using MF = MFMA32<DT>;
  using frag = typename MF::frag;
  using P4 = typename Pack4<DT>::V;
  constexpr int QT = 32;
  constexpr int QS = QT + 8;
  constexpr int DS = D + 8;
  constexpr int NKC = D / 16;
  constexpr int NMT = D / 32;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  T* Q_lds = (T*)smem_raw;                   // [2][QT][DS]
  T* dO_lds = Q_lds + 2 * QT * DS;           // [2][QT][DS]
  T* PT_lds = dO_lds + 2 * QT * DS;          // [4][32][QS] (P^T or dS^T)
  float* lse_lds = (float*)(PT_lds + 4 * 32 * QS);   // [2][QT]
  float* dl_lds = lse_lds + 2 * QT;                  // [2][QT] (dk only)

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo32 = lane & 31;
  const int hi5 = lane >> 5;

  const int nKT = (S + 127) / 128;
  int bid = blockIdx.x;
  const int kt = bid % nKT;
  const int h = (bid / nKT) % Hq;
  const int b = bid / (nKT * Hq);
  const int hkv = h / (Hq / Hkv);

  const int k0 = kt * 128 + wave * 32;
  const int64_t qoff = (((int64_t)b * Hq + h) * S) * D;
  const int64_t kvoff = (((int64_t)b * Hkv + hkv) * S) * D;
  const int64_t lseoff = ((int64_t)b * Hq + h) * S;
  const int64_t voff = v_sb ? ((int64_t)b * v_sb + (int64_t)hkv * v_sh) : kvoff;
  const int64_t v_rs = v_sb ? v_sr : (int64_t)D;

  // K (and V for dk) B-fragment base: re-read from L2 per q-tile, freeing
  // the register copies (v3-port trick)
  const int kr_c0 = (k0 + lo32) < S ? (k0 + lo32) : S - 1;
  const T* kbase_p = k + kvoff + (int64_t)kr_c0 * D + hi5 * 8;
  const T* vbase_p = v + voff + (int64_t)kr_c0 * v_rs + hi5 * 8;

  floatx16 acc[NMT];
#pragma unroll
  for (int mt = 0; mt < NMT; ++mt) acc[mt] = (floatx16)(0.f);

  const int qstart = (kt * 128) / QT;
  const int nQT2 = (S + QT - 1) / QT;

  const int st_t = (int)threadIdx.x;
  const bool st_on = st_t < (QT * D) / 8;
  const int st_row = st_t / (D / 8);
  const int st_c8 = (st_t % (D / 8)) * 8;
  shortx8 qreg, dreg;
  float lse_reg = 0.f, dl_reg = 0.f;
  const int64_t gbase = (int64_t)b * g_sb + (int64_t)h * g_sh;

  auto load_qtile = [&](int qt2) {
    const int qrow = qt2 * QT + st_row;
    const int qr_c = qrow < S ? qrow : S - 1;
    if (st_on) {
      qreg = *(const shortx8*)(q + qoff + (int64_t)qr_c * D + st_c8);
      dreg = *(const shortx8*)(do_ + gbase + (int64_t)qr_c * g_sr + st_c8);
    }
    if (st_t < QT) {
      const int rr = qt2 * QT + st_t;
      const int rr_c = rr < S ? rr : S - 1;
      lse_reg = lse[lseoff + rr_c];
      if (WANT_DK) dl_reg = delta[lseoff + rr_c];
    }
  };
  auto write_qtile = [&](int buf) {
    if (st_on) {
      *(shortx8*)(Q_lds + buf * QT * DS + st_row * DS + st_c8) = qreg;
      *(shortx8*)(dO_lds + buf * QT * DS + st_row * DS + st_c8) = dreg;
    }
    if (st_t < QT) {
      lse_lds[buf * QT + st_t] = lse_reg;
      if (WANT_DK) dl_lds[buf * QT + st_t] = dl_reg;
    }
  };

  load_qtile(qstart);
  write_qtile(0);
  __syncthreads();

  for (int qt2 = qstart; qt2 < nQT2; ++qt2) {
    const int qbase = qt2 * QT;
    const int cur = (qt2 - qstart) & 1;
    T* Qb = Q_lds + cur * QT * DS;
    T* dOb = dO_lds + cur * QT * DS;
    const float* lse_b = lse_lds + cur * QT;
    const float* dl_b = dl_lds + cur * QT;
    if (qt2 + 1 < nQT2) load_qtile(qt2 + 1);

    // ---- S (and dP for dk) TRANSPOSED-C: C col = key = lo32,
    //      row = q = (r&3) + 8*(r>>2) + 4*hi5 ----
    floatx16 st = (floatx16)(0.f), dpt = (floatx16)(0.f);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kc = 0; kc < NKC; ++kc) {
      frag qa = *(const frag*)(Qb + lo32 * DS + kc * 16 + hi5 * 8);
      frag kb = *(const frag*)(kbase_p + kc * 16);
      st = MF::mma(qa, kb, st);
      if (WANT_DK) {
        frag doa = *(const frag*)(dOb + lo32 * DS + kc * 16 + hi5 * 8);
        frag vb = *(const frag*)(vbase_p + kc * 16);
        dpt = MF::mma(doa, vb, dpt);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    const int krow = k0 + lo32;
    const bool full = (qbase >= k0 + 32) && (qbase + QT <= S);
    // pack P^T (dv) or dS^T (dk) into the [32 key][QS] row-major image,
    // one ds_write_b64 per 4 consecutive q
    T* Pw = PT_lds + wave * 32 * QS;
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const int qloc = 8 * g + 4 * hi5;
      const floatx4 lse4 = *(const floatx4*)(lse_b + qloc);
      floatx4 dl4;
      if (WANT_DK) dl4 = *(const floatx4*)(dl_b + qloc);
      P4 pk;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int i = g * 4 + r;
        float pv;
        if (full) {
          pv = __expf(st[i] * scale - lse4[r]);
        } else {
          const int qrow = qbase + qloc + r;
          pv = 0.f;
          if (krow <= qrow && krow < S && qrow < S)
            pv = __expf(st[i] * scale - lse4[r]);
        }
        if (WANT_DK) pv = pv * (dpt[i] - dl4[r]) * scale;
        ((T*)&pk)[r] = TR::fromF(pv);
      }
      *(P4*)(Pw + lo32 * QS + qloc) = pk;
    }

    // ---- acc += A(P^T|dS^T) x B(dO_T|Q_T) ----
    frag pa[2];
#pragma unroll
    for (int c = 0; c < 2; ++c)
      pa[c] = *(const frag*)(Pw + lo32 * QS + c * 16 + hi5 * 8);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mt = 0; mt < NMT; ++mt)
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        frag bb = trread_afrag32<DT>(WANT_DK ? Qb : dOb, c * 16, mt * 32, DS, lane);
        acc[mt] = MF::mma(pa[c], bb, acc[mt]);
      }
    __builtin_amdgcn_s_setprio(0);

    if (qt2 + 1 < nQT2) write_qtile(cur ^ 1);
    __syncthreads();
  }

  // ---- write dV / dK (per q-head; caller sums GQA groups) ----
  const int64_t obase = o_sb ? ((int64_t)b * o_sb + (int64_t)h * o_sh) : qoff;
  const int64_t o_rs = o_sb ? o_sr : (int64_t)D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kr = k0 + (r & 3) + 8 * (r >> 2) + 4 * hi5;
    if (kr >= S) continue;
#pragma unroll
    for (int mt = 0; mt < NMT; ++mt)
      out[obase + (int64_t)kr * o_rs + mt * 32 + lo32] = TR::fromF(acc[mt][r]);
  }
}

template <int DT, int D>
__global__ __launch_bounds__(256) void attn_bwd_dq_v3_kernel(
    typename DTraits<DT>::T* __restrict__ dq_out,
    const typename DTraits<DT>::T* __restrict__ do_,
    const typename DTraits<DT>::T* __restrict__ q,
    const typename DTraits<DT>::T* __restrict__ k,
    const typename DTraits<DT>::T* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    int B, int Hq, int Hkv, int S, float scale,
    int64_t g_sb, int64_t g_sh, int64_t g_sr) {
  using TR = DTraits<DT>;
  using T = typename TR::T;
  using MF = MFMA32<DT>;
  using frag = typename MF::frag;
  constexpr int KT = 32;
  constexpr int KS = KT + 8;
  constexpr int DS = D + 8;
  constexpr int NKC = D / 16;
  constexpr int NMT = D / 32;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  T* K_lds = (T*)smem_raw;                    // [2][KT][DS]
  T* V_lds = K_lds + 2 * KT * DS;             // [2][KT][DS]
  T* S_lds = V_lds + 2 * KT * DS;             // [4][32][KS]

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int lo32 = lane & 31;
  const int hi5 = lane >> 5;

  const int nQT = (S + 127) / 128;
  int bid = blockIdx.x;
  const int qt = bid % nQT;
  const int h = (bid / nQT) % Hq;
  const int b = bid / (nQT * Hq);
  const int hkv = h / (Hq / Hkv);

  const int q0 = qt * 128 + wave * 32;
  const int64_t qoff = (((int64_t)b * Hq + h) * S) * D;
  const int64_t kvoff = (((int64_t)b * Hkv + hkv) * S) * D;
  const int64_t lseoff = ((int64_t)b * Hq + h) * S;

  // A-fragments (row = q = lane&31) + per-C-row lse/delta
  frag q_frag[NKC], do_frag[NKC];
  float lse_r[16], dl_r[16];
  {
    const int qrow = q0 + lo32;
    const int qr_c = qrow < S ? qrow : S - 1;
    const int64_t gb = (int64_t)b * g_sb + (int64_t)h * g_sh;
#pragma unroll
    for (int kc = 0; kc < NKC; ++kc) {
      q_frag[kc] = *(const frag*)(q + qoff + (int64_t)qr_c * D + kc * 16 + hi5 * 8);
      do_frag[kc] = *(const frag*)(do_ + gb + (int64_t)qr_c * g_sr + kc * 16 + hi5 * 8);
    }
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int rr = q0 + (r & 3) + 8 * (r >> 2) + 4 * hi5;
      const int rr_c = rr < S ? rr : S - 1;
      lse_r[r] = lse[lseoff + rr_c];
      dl_r[r] = delta[lseoff + rr_c];
    }
  }

  floatx16 dq_acc[NMT];
#pragma unroll
  for (int mt = 0; mt < NMT; ++mt) dq_acc[mt] = (floatx16)(0.f);

  const int kv_end = min(S, qt * 128 + 128);
  const int n_kt = (kv_end + KT - 1) / KT;

  const int st_t = (int)threadIdx.x;
  const bool st_on = st_t < (KT * D) / 8;
  const int st_row = st_t / (D / 8);
  const int st_c8 = (st_t % (D / 8)) * 8;
  shortx8 kreg, vreg;

  auto load_ktile = [&](int kt2) {
    if (st_on) {
      const int krow = kt2 * KT + st_row;
      const int kr_c = krow < S ? krow : S - 1;
      kreg = *(const shortx8*)(k + kvoff + (int64_t)kr_c * D + st_c8);
      vreg = *(const shortx8*)(v + kvoff + (int64_t)kr_c * D + st_c8);
    }
  };
  auto write_ktile = [&](int buf) {
    if (st_on) {
      *(shortx8*)(K_lds + buf * KT * DS + st_row * DS + st_c8) = kreg;
      *(shortx8*)(V_lds + buf * KT * DS + st_row * DS + st_c8) = vreg;
    }
  };

  load_ktile(0);
  write_ktile(0);
  __syncthreads();

  for (int kt2 = 0; kt2 < n_kt; ++kt2) {
    const int kbase = kt2 * KT;
    const int cur = kt2 & 1;
    T* Kb = K_lds + cur * KT * DS;
    T* Vb = V_lds + cur * KT * DS;
    if (kt2 + 1 < n_kt) load_ktile(kt2 + 1);

    // ---- S = Q K^T and dP = dO V^T  (C: col = key = lane&31) ----
    floatx16 st = (floatx16)(0.f), dpt = (floatx16)(0.f);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kc = 0; kc < NKC; ++kc) {
      frag bk = *(const frag*)(Kb + lo32 * DS + kc * 16 + hi5 * 8);
      st = MF::mma(q_frag[kc], bk, st);
      frag bv = *(const frag*)(Vb + lo32 * DS + kc * 16 + hi5 * 8);
      dpt = MF::mma(do_frag[kc], bv, dpt);
    }
    __builtin_amdgcn_s_setprio(0);

    const int kcol = kbase + lo32;
    const bool full = (kbase + KT <= q0) && (kbase + KT <= S) && (q0 + 32 <= S);
    float ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      float pv;
      if (full) {
        pv = __expf(st[r] * scale - lse_r[r]);
      } else {
        const int qrow = q0 + (r & 3) + 8 * (r >> 2) + 4 * hi5;
        pv = 0.f;
        if (kcol <= qrow && kcol < S && qrow < S)
          pv = __expf(st[r] * scale - lse_r[r]);
      }
      ds[r] = pv * (dpt[r] - dl_r[r]) * scale;
    }

    // ---- dQ += dS K (A = dS via LDS; B = K_T) ----
    T* Sw = S_lds + wave * 32 * KS;
#pragma unroll
    for (int r = 0; r < 16; ++r)
      Sw[((r & 3) + 8 * (r >> 2) + 4 * hi5) * KS + lo32] = TR::fromF(ds[r]);
    frag da[2];
#pragma unroll
    for (int c = 0; c < 2; ++c)
      da[c] = *(const frag*)(Sw + lo32 * KS + c * 16 + hi5 * 8);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mt = 0; mt < NMT; ++mt)
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        frag bk = trread_afrag32<DT>(Kb, c * 16, mt * 32, DS, lane);
        dq_acc[mt] = MF::mma(da[c], bk, dq_acc[mt]);
      }
    __builtin_amdgcn_s_setprio(0);

    if (kt2 + 1 < n_kt) write_ktile(cur ^ 1);
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qrow = q0 + (r & 3) + 8 * (r >> 2) + 4 * hi5;
    if (qrow >= S) continue;
#pragma unroll
    for (int mt = 0; mt < NMT; ++mt)
      dq_out[qoff + (int64_t)qrow * D + mt * 32 + lo32] = TR::fromF(dq_acc[mt][r]);
  }
}

// ======================= C-ABI wrappers =======================

static bool use_attn_v2() {
  static int cached = -1;
  if (cached < 0) {
    const char* e = getenv("DK_ATTN_V2");
    cached = (e && e[0] == '1') ? 1 : 0;
  }
  return cached == 1;
}

// backward defaults to the 16x16 (v2) kernels: measured same-box A/B has
// them ~7% faster than the 32x32 port (139 vs 148 TF); DK_ATTN_BWD_V3=1
// opts into the v3 backward for future tuning.
static bool use_bwd_split32() {
  // split dv-only/dk-only 32x32 backward (A/B vs the fused 16x16 pair)
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("DK_ATTN_BWD_SPLIT32");
    v = (e && e[0] == '1') ? 1 : 0;
  }
  return v != 0;
}

static bool use_bwd_v3() {
  static int cached = -1;
  if (cached < 0) {
    const char* e = getenv("DK_ATTN_BWD_V3");
    cached = (e && e[0] == '1') ? 1 : 0;
  }
  return cached == 1;
}

template <int DT, int D>
static int launch_attn_fwd(void* o, float* lse, const void* q, const void* k,
                           const void* v, int64_t B, int64_t Hq, int64_t Hkv,
                           int64_t S, float scale, int64_t o_sb, int64_t o_sh,
                           int64_t o_sr, int64_t v_sb, int64_t v_sh, int64_t v_sr,
                           dkStream stream) {
  using T = typename DTraits<DT>::T;
  constexpr int KT = 64, KS = KT + 8, DS = D + 8;
  if (!use_attn_v2()) {
    // v3: swapped QK^T, in-register softmax, 128 q rows per workgroup
    const int nQT = (int)((S + 127) / 128);
    const int grid = (int)(B * Hq * nQT);
    const size_t lds = sizeof(T) * (2 * KT * DS + 2 * KT * DS);
    hipLaunchKernelGGL((attn_fwd_v3_kernel<DT, D>), dim3(grid), dim3(256), lds,
                       (hipStream_t)stream, (T*)o, lse, (const T*)q, (const T*)k,
                       (const T*)v, (int)B, (int)Hq, (int)Hkv, (int)S, scale,
                       o_sb, o_sh, o_sr, v_sb, v_sh, v_sr);
    DK_CHECK_LAUNCH();
    return 0;
  }
  const int nQT = (int)((S + 63) / 64);
  const int grid = (int)(B * Hq * nQT);
  const size_t lds = sizeof(T) * (2 * KT * DS + 2 * KT * DS + 4 * 16 * KS);
  hipLaunchKernelGGL((attn_fwd_kernel<DT, D>), dim3(grid), dim3(256), lds,
                     (hipStream_t)stream, (T*)o, lse, (const T*)q, (const T*)k,
                     (const T*)v, (int)B, (int)Hq, (int)Hkv, (int)S, scale,
                     o_sb, o_sh, o_sr, v_sb, v_sh, v_sr);
  DK_CHECK_LAUNCH();
  return 0;
}

extern "C" int dk_attn_fwd(void* o, float* lse, const void* q, const void* k,
                           const void* v, int64_t B, int64_t Hq, int64_t Hkv,
                           int64_t S, int64_t D, float scale,
                           int64_t o_sb, int64_t o_sh, int64_t o_sr,
                           int64_t v_sb, int64_t v_sh, int64_t v_sr, int dtype,
                           dkStream stream) {
  if (dtype != 1 && dtype != 2) return (int)hipErrorInvalidValue;
  if (o_sb == 0) { o_sb = Hq * S * D; o_sh = S * D; o_sr = D; }  // BHSD default
  if (D == 64) {
    if (dtype == 2) return launch_attn_fwd<2, 64>(o, lse, q, k, v, B, Hq, Hkv, S, scale, o_sb, o_sh, o_sr, v_sb, v_sh, v_sr, stream);
    return launch_attn_fwd<1, 64>(o, lse, q, k, v, B, Hq, Hkv, S, scale, o_sb, o_sh, o_sr, v_sb, v_sh, v_sr, stream);
  } else if (D == 32) {
    if (dtype == 2) return launch_attn_fwd<2, 32>(o, lse, q, k, v, B, Hq, Hkv, S, scale, o_sb, o_sh, o_sr, v_sb, v_sh, v_sr, stream);
    return launch_attn_fwd<1, 32>(o, lse, q, k, v, B, Hq, Hkv, S, scale, o_sb, o_sh, o_sr, v_sb, v_sh, v_sr, stream);
  }
  return (int)hipErrorInvalidValue;
}

extern "C" int dk_attn_bwd_preprocess(float* delta, const void* do_, const void* o,
                                      int64_t B, int64_t H, int64_t S, int64_t D,
                                      int64_t g_sb, int64_t g_sh, int64_t g_sr,
                                      int dtype, dkStream stream) {
  const int64_t rows = B * H * S;
  if (g_sb == 0) { g_sb = H * S * D; g_sh = S * D; g_sr = D; }
  int64_t g = (rows + 3) / 4;  // 4 waves per block, one row per wave-iteration
  int grid = (int)(g > 2048 ? 2048 : (g < 1 ? 1 : g));
  DK_DISPATCH_DT(dtype, {
    if constexpr (kDT != 0) {
      using T = typename DTraits<kDT>::T;
      hipLaunchKernelGGL((attn_bwd_pre_kernel<kDT>), dim3(grid), dim3(256), 0,
                         (hipStream_t)stream, delta, (const T*)do_, (const T*)o,
                         rows, (int)H, (int)S, (int)D, g_sb, g_sh, g_sr);
    } else {
      return (int)hipErrorInvalidValue;
    }
  });
  DK_CHECK_LAUNCH();
  return 0;
}

template <int DT, int D>
static int launch_attn_bwd_dkdv(void* dk_o, void* dv_o, const void* do_, const void* q,
                                const void* k, const void* v, const float* lse,
                                const float* delta, int64_t B, int64_t Hq, int64_t Hkv,
                                int64_t S, float scale, int64_t g_sb, int64_t g_sh,
                                int64_t g_sr, int64_t v_sb, int64_t v_sh, int64_t v_sr,
                                int64_t dv_sb, int64_t dv_sh, int64_t dv_sr,
                                dkStream stream) {
  using T = typename DTraits<DT>::T;
  constexpr int QT = 32, QS = QT + 8, DS = D + 8;
  if (use_bwd_split32()) {
    const int nKTs = (int)((S + 127) / 128);
    const int grids = (int)(B * Hq * nKTs);
    const size_t ldss = sizeof(T) * (4 * QT * DS + 4 * 32 * QS) + sizeof(float) * 4 * QT;
    hipLaunchKernelGGL((attn_bwd_split32_kernel<DT, D, 0>), dim3(grids), dim3(256), ldss,
                       (hipStream_t)stream, (T*)dv_o, (const T*)do_, (const T*)q,
                       (const T*)k, (const T*)v, lse, delta, (int)B, (int)Hq,
                       (int)Hkv, (int)S, scale, g_sb, g_sh, g_sr, v_sb, v_sh, v_sr,
                       dv_sb, dv_sh, dv_sr);
    DK_CHECK_LAUNCH();
    hipLaunchKernelGGL((attn_bwd_split32_kernel<DT, D, 1>), dim3(grids), dim3(256), ldss,
                       (hipStream_t)stream, (T*)dk_o, (const T*)do_, (const T*)q,
                       (const T*)k, (const T*)v, lse, delta, (int)B, (int)Hq,
                       (int)Hkv, (int)S, scale, g_sb, g_sh, g_sr, v_sb, v_sh, v_sr,
                       0, 0, 0);
    DK_CHECK_LAUNCH();
    return 0;
  }
  if (use_bwd_v3() && v_sb == 0 && dv_sb == 0) {  // v3 port is contiguous-only
    const int nKT3 = (int)((S + 127) / 128);
    const int grid3 = (int)(B * Hq * nKT3);
    const size_t lds3 = sizeof(T) * (4 * QT * DS + 2 * 4 * 32 * QS)
                        + sizeof(float) * 4 * QT;
    hipLaunchKernelGGL((attn_bwd_dkdv_v3_kernel<DT, D>), dim3(grid3), dim3(256), lds3,
                       (hipStream_t)stream, (T*)dk_o, (T*)dv_o, (const T*)do_,
                       (const T*)q, (const T*)k, (const T*)v, lse, delta,
                       (int)B, (int)Hq, (int)Hkv, (int)S, scale, g_sb, g_sh, g_sr);
    DK_CHECK_LAUNCH();
    return 0;
  }
  const int nKT = (int)((S + 127) / 128);    // 8-wave WG: 128 keys
  const int grid = (int)(B * Hq * nKT);
  constexpr int QTT = 64;  // staged q tile (2 compute halves per barrier)
  const size_t lds = sizeof(T) * (4 * QTT * DS + 2 * 8 * 16 * QS) + sizeof(float) * 4 * QTT;
  hipLaunchKernelGGL((attn_bwd_dkdv_kernel<DT, D>), dim3(grid), dim3(512), lds,
                     (hipStream_t)stream, (T*)dk_o, (T*)dv_o, (const T*)do_,
                     (const T*)q, (const T*)k, (const T*)v, lse, delta,
                     (int)B, (int)Hq, (int)Hkv, (int)S, scale, g_sb, g_sh, g_sr,
                     v_sb, v_sh, v_sr, dv_sb, dv_sh, dv_sr);
  DK_CHECK_LAUNCH();
  return 0;
}

extern "C" int dk_attn_bwd_dkdv(void* dk_o, void* dv_o, const void* do_, const void* q,
                                const void* k, const void* v, const float* lse,
                                const float* delta, int64_t B, int64_t Hq, int64_t Hkv,
                                int64_t S, int64_t D, float scale,
                                int64_t g_sb, int64_t g_sh, int64_t g_sr,
                                int64_t v_sb, int64_t v_sh, int64_t v_sr,
                                int64_t dv_sb, int64_t dv_sh, int64_t dv_sr, int dtype,
                                dkStream stream) {
  if (dtype != 1 && dtype != 2) return (int)hipErrorInvalidValue;
  if (g_sb == 0) { g_sb = Hq * S * D; g_sh = S * D; g_sr = D; }
  if (D == 64) {
    if (dtype == 2) return launch_attn_bwd_dkdv<2, 64>(dk_o, dv_o, do_, q, k, v, lse, delta, B, Hq, Hkv, S, scale, g_sb, g_sh, g_sr, v_sb, v_sh, v_sr, dv_sb, dv_sh, dv_sr, stream);
    return launch_attn_bwd_dkdv<1, 64>(dk_o, dv_o, do_, q, k, v, lse, delta, B, Hq, Hkv, S, scale, g_sb, g_sh, g_sr, v_sb, v_sh, v_sr, dv_sb, dv_sh, dv_sr, stream);
  } else if (D == 32) {
    if (dtype == 2) return launch_attn_bwd_dkdv<2, 32>(dk_o, dv_o, do_, q, k, v, lse, delta, B, Hq, Hkv, S, scale, g_sb, g_sh, g_sr, v_sb, v_sh, v_sr, dv_sb, dv_sh, dv_sr, stream);
    return launch_attn_bwd_dkdv<1, 32>(dk_o, dv_o, do_, q, k, v, lse, delta, B, Hq, Hkv, S, scale, g_sb, g_sh, g_sr, v_sb, v_sh, v_sr, dv_sb, dv_sh, dv_sr, stream);
  }
  return (int)hipErrorInvalidValue;
}

template <int DT, int D>
static int launch_attn_bwd_dq(void* dq_o, const void* do_, const void* q, const void* k,
                              const void* v, const float* lse, const float* delta,
                              int64_t B, int64_t Hq, int64_t Hkv, int64_t S,
                              float scale, int64_t g_sb, int64_t g_sh, int64_t g_sr,
                              int64_t v_sb, int64_t v_sh, int64_t v_sr,
                              dkStream stream) {
  using T = typename DTraits<DT>::T;
  constexpr int KT = 32, KS = KT + 8, DS = D + 8;   // v3 tile constants
  constexpr int KT2 = 64, KS2 = KT2 + 8;            // v2 8-wave tile constants
  if (use_bwd_v3() && v_sb == 0) {  // v3 port is contiguous-only
    const int nQT3 = (int)((S + 127) / 128);
    const int grid3 = (int)(B * Hq * nQT3);
    const size_t lds3 = sizeof(T) * (4 * KT * DS + 4 * 32 * KS);
    hipLaunchKernelGGL((attn_bwd_dq_v3_kernel<DT, D>), dim3(grid3), dim3(256), lds3,
                       (hipStream_t)stream, (T*)dq_o, (const T*)do_, (const T*)q,
                       (const T*)k, (const T*)v, lse, delta, (int)B, (int)Hq,
                       (int)Hkv, (int)S, scale, g_sb, g_sh, g_sr);
    DK_CHECK_LAUNCH();
    return 0;
  }
  const int nQT = (int)((S + 127) / 128);    // 8-wave WG: 128 q rows
  const int grid = (int)(B * Hq * nQT);
  const size_t lds = sizeof(T) * (6 * KT2 * DS + 8 * 16 * KS2);  // 3-ring K+V
  hipLaunchKernelGGL((attn_bwd_dq_kernel<DT, D>), dim3(grid), dim3(512), lds,
                     (hipStream_t)stream, (T*)dq_o, (const T*)do_, (const T*)q,
                     (const T*)k, (const T*)v, lse, delta, (int)B, (int)Hq,
                     (int)Hkv, (int)S, scale, g_sb, g_sh, g_sr, v_sb, v_sh, v_sr);
  DK_CHECK_LAUNCH();
  return 0;
}

extern "C" int dk_attn_bwd_dq(void* dq_o, const void* do_, const void* q, const void* k,
                              const void* v, const float* lse, const float* delta,
                              int64_t B, int64_t Hq, int64_t Hkv, int64_t S, int64_t D,
                              float scale, int64_t g_sb, int64_t g_sh, int64_t g_sr,
                              int64_t v_sb, int64_t v_sh, int64_t v_sr,
                              int dtype, dkStream stream) {
  if (dtype != 1 && dtype != 2) return (int)hipErrorInvalidValue;
  if (g_sb == 0) { g_sb = Hq * S * D; g_sh = S * D; g_sr = D; }
  if (D == 64) {
    if (dtype == 2) return launch_attn_bwd_dq<2, 64>(dq_o, do_, q, k, v, lse, delta, B, Hq, Hkv, S, scale, g_sb, g_sh, g_sr, v_sb, v_sh, v_sr, stream);
    return launch_attn_bwd_dq<1, 64>(dq_o, do_, q, k, v, lse, delta, B, Hq, Hkv, S, scale, g_sb, g_sh, g_sr, v_sb, v_sh, v_sr, stream);
  } else if (D == 32) {
    if (dtype == 2) return launch_attn_bwd_dq<2, 32>(dq_o, do_, q, k, v, lse, delta, B, Hq, Hkv, S, scale, g_sb, g_sh, g_sr, v_sb, v_sh, v_sr, stream);
    return launch_attn_bwd_dq<1, 32>(dq_o, do_, q, k, v, lse, delta, B, Hq, Hkv, S, scale, g_sb, g_sh, g_sr, v_sb, v_sh, v_sr, stream);
  }
  return (int)hipErrorInvalidValue;
}

// ======================= MFMA layout probe (test-only) =======================
// D = A[16][32] x B[32][16] using the fragment maps above; out is [16][16]
// f32 row-major.  A/B given row-major bf16.  The GPU test feeds asymmetric
// random matrices and compares with a host matmul (guide §5.4 rule 16).
__global__ void probe_mfma_kernel(float* __restrict__ out,
                                  const unsigned short* __restrict__ a,
                                  const unsigned short* __restrict__ b) {
  const int lane = threadIdx.x & 63;
  const int lo = lane & 15, hi = lane >> 4;
  shortx8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    ((unsigned short*)&af)[j] = a[lo * 32 + hi * 8 + j];        // A[row=lo][k=8*hi+j]
    ((unsigned short*)&bf)[j] = b[(hi * 8 + j) * 16 + lo];      // B[k=8*hi+j][col=lo]
  }
  floatx4 c = (floatx4)(0.f);
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) out[(hi * 4 + r) * 16 + lo] = c[r];  // C[row=(hi*4+r)][col=lo]
}

extern "C" int dk_probe_mfma_16x16x32_bf16(float* out_d, const void* a16x32,
                                           const void* b32x16, dkStream stream) {
  hipLaunchKernelGGL(probe_mfma_kernel, dim3(1), dim3(64), 0, (hipStream_t)stream,
                     out_d, (const unsigned short*)a16x32, (const unsigned short*)b32x16);
  DK_CHECK_LAUNCH();
  return 0;
}

// Alternative candidate mapping (k interleaved: k = (l>>4) + 4*j) — a
// diagnostic twin so one GPU round can identify the true layout if the
// primary assumption fails.
__global__ void probe_mfma_alt_kernel(float* __restrict__ out,
                                      const unsigned short* __restrict__ a,
                                      const unsigned short* __restrict__ b) {
  const int lane = threadIdx.x & 63;
  const int lo = lane & 15, hi = lane >> 4;
  shortx8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    ((unsigned short*)&af)[j] = a[lo * 32 + (hi + 4 * j)];
    ((unsigned short*)&bf)[j] = b[(hi + 4 * j) * 16 + lo];
  }
  floatx4 c = (floatx4)(0.f);
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) out[(hi * 4 + r) * 16 + lo] = c[r];
}

extern "C" int dk_probe_mfma_16x16x32_bf16_alt(float* out_d, const void* a16x32,
                                               const void* b32x16, dkStream stream) {
  hipLaunchKernelGGL(probe_mfma_alt_kernel, dim3(1), dim3(64), 0, (hipStream_t)stream,
                     out_d, (const unsigned short*)a16x32, (const unsigned short*)b32x16);
  DK_CHECK_LAUNCH();
  return 0;
}

// 32x32x16 bf16 probe — assumed maps:
//   A[32][16]: lane holds row = l&31, k = 8*(l>>5) + j (j = 0..7)
//   B[16][32]: lane holds col = l&31, k = 8*(l>>5) + j
//   C[32][32]: lane holds col = l&31, row = (r&3) + 8*(r>>2) + 4*(l>>5)
typedef __attribute__((ext_vector_type(16))) float floatx16;
__global__ void probe_mfma32_kernel(float* __restrict__ out,
                                    const unsigned short* __restrict__ a,
                                    const unsigned short* __restrict__ b) {
  const int lane = threadIdx.x & 63;
  const int lo = lane & 31, hi = lane >> 5;
  shortx8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    ((unsigned short*)&af)[j] = a[lo * 16 + hi * 8 + j];
    ((unsigned short*)&bf)[j] = b[(hi * 8 + j) * 32 + lo];
  }
  floatx16 c = (floatx16)(0.f);
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r)
    out[((r & 3) + 8 * (r >> 2) + 4 * hi) * 32 + lo] = c[r];
}

extern "C" int dk_probe_mfma_32x32x16_bf16(float* out_d, const void* a32x16,
                                           const void* b16x32, dkStream stream) {
  hipLaunchKernelGGL(probe_mfma32_kernel, dim3(1), dim3(64), 0, (hipStream_t)stream,
                     out_d, (const unsigned short*)a32x16, (const unsigned short*)b16x32);
  DK_CHECK_LAUNCH();
  return 0;
}

// permlane32_swap semantics probe: in a[lane] = lane, b[lane] = 100+lane;
// writes the two results so the host can read the exact lane exchange.
__global__ void probe_permlane_kernel(int* __restrict__ out) {
  const int lane = threadIdx.x & 63;
  int a = lane, b = 100 + lane;
  auto r = __builtin_amdgcn_permlane32_swap(a, b, false, false);
  out[lane] = r[0];
  out[64 + lane] = r[1];
}

extern "C" int dk_probe_permlane32(int* out_d, dkStream stream) {
  hipLaunchKernelGGL(probe_permlane_kernel, dim3(1), dim3(64), 0, (hipStream_t)stream, out_d);
  DK_CHECK_LAUNCH();
  return 0;
}
