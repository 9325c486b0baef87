// dtmx MFMA GEMM core + implicit-GEMM convolution for gfx950 (MI355X).
//
// One TN GEMM template serves FullyConnected fwd/dgrad/wgrad, conv fwd
// (im2col gather on the A operand), conv dgrad (dy gather) and conv wgrad
// (transposed operands + split-K) — the reference's cuBLAS/cuDNN/im2col stack
// (src/operator/nn/convolution.cu, fully_connected-inl.h, linalg_impl.h)
// redesigned as CDNA4 MFMA kernels.
//
// Structure (cdna_hip_programming.md §5 ladder step 3):
//   - 128x128 block tile, BK=64, 256 threads (4 waves, 2x2 wave grid,
//     64x64 per wave = 4x4 fragments of v_mfma_f32_16x16x32_bf16)
//   - global->LDS staging via 16-B global_load_lds (async, no VGPR round trip)
//   - LDS images [128 rows][64 k] bf16, XOR-swizzled byte^=((row&7)<<4) on the
//     per-lane *source* address (rule 21) against ds_read_b128 bank conflicts
//   - double-buffered; one vmcnt(0)+barrier per K-tile; s_setprio around MFMA
//   - XCD-bijective block swizzle for L2 locality
// fp32 accumulation; bf16 (or fp32-atomic split-K) epilogues.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <map>
#include <tuple>

#include "dtmx_common.h"

namespace dtmx {

// ---------------------------------------------------------------- providers
// A provider returns the global address of the 16-B piece holding bf16
// elements [k8*8, k8*8+8) of logical row m, or `zero` when out of range.

// Providers expose a per-row context (Row) computed ONCE per block — the
// block's rows are fixed across the whole K loop, so the (n,p,q)/(n,h,w)
// FastDiv decode does not belong in the per-piece staging path (measured
// gather overhead vs same-shape dense GEMM ~30-40%).
template <typename elem_t>
struct DenseP {
  using elem = elem_t;
  static constexpr bool kDense = true;
  const elem_t* base;
  const elem_t* zero;
  uint32_t M, K;  // rows, k extent (elements, multiple of 8)
  uint32_t ld;    // row stride in elements
  struct Row {
    const elem_t* p;  // nullptr = out-of-range row
  };
  __device__ __forceinline__ Row row(uint32_t m) const {
    return {m < M ? base + (size_t)m * ld : nullptr};
  }
  __device__ __forceinline__ const void* addr(const Row& r, uint32_t k8) const {
    uint32_t k = k8 * 8;
    if (!r.p || k >= K) return zero;
    return r.p + k;
  }
};

// conv forward A: row m = (n,p,q) output pixel, k = (r,s,c), x is NHWC.
template <typename elem_t>
struct ConvFwdA {
  using elem = elem_t;
  static constexpr bool kDense = false;
  const elem_t* x;
  const elem_t* zero;
  uint32_t M, Ktot;              // M = N*P*Q, Ktot = R*S*C
  uint32_t C, H, W, Q, S;
  int u, v, ph, pw;              // stride, padding
  FastDiv dQ, dPQ, dC, dS;
  struct Row {
    const elem_t* pixel0;  // &x[n][p*u-ph][q*v-pw][0] (may point out of range)
    int ih0, iw0;          // p*u-ph, q*v-pw
    uint32_t valid;
  };
  __device__ __forceinline__ Row row(uint32_t m) const {
    Row r{};
    if (m >= M) return r;
    uint32_t n = dPQ.div(m), pq = dPQ.mod(m, n);
    uint32_t p = dQ.div(pq), q = dQ.mod(pq, p);
    r.ih0 = (int)(p * u) - ph;
    r.iw0 = (int)(q * v) - pw;
    r.pixel0 = x + (((int64_t)n * H + r.ih0) * W + r.iw0) * (int64_t)C;
    r.valid = 1;
    return r;
  }
  __device__ __forceinline__ const void* addr(const Row& rc, uint32_t k8) const {
    uint32_t k = k8 * 8;
    if (!rc.valid || k >= Ktot) return zero;
    uint32_t rs = dC.div(k), c = dC.mod(k, rs);
    uint32_t r = dS.div(rs), s = dS.mod(rs, r);
    if ((uint32_t)(rc.ih0 + (int)r) >= H || (uint32_t)(rc.iw0 + (int)s) >= W)
      return zero;
    return rc.pixel0 + ((int64_t)r * W + s) * C + c;
  }
};

// conv dgrad A: row m = (n,h,w) input pixel, k = (r,s,ko), dy is NHWC.
// dx[n,h,w,c] = sum_{r,s,ko} dy[n,(h+ph-r)/u,(w+pw-s)/v,ko] * w[ko,r,s,c]
template <typename elem_t>
struct ConvDgradA {
  using elem = elem_t;
  static constexpr bool kDense = false;
  const elem_t* dy;
  const elem_t* zero;
  uint32_t M, Ktot;              // M = N*H*W, Ktot = R*S*Kout
  uint32_t Ko, H, W, P, Q, S;
  int u, v, ph, pw;
  FastDiv dW_, dHW, dKo, dS;
  struct Row {
    const elem_t* base_n;  // &dy[n][0][0][0]
    int hp, wp;            // h+ph, w+pw
    uint32_t valid;
  };
  __device__ __forceinline__ Row row(uint32_t m) const {
    Row r{};
    if (m >= M) return r;
    uint32_t n = dHW.div(m), hw = dHW.mod(m, n);
    uint32_t h = dW_.div(hw), w = dW_.mod(hw, h);
    r.hp = (int)h + ph;
    r.wp = (int)w + pw;
    r.base_n = dy + (size_t)n * P * Q * Ko;
    r.valid = 1;
    return r;
  }
  __device__ __forceinline__ const void* addr(const Row& rc, uint32_t k8) const {
    uint32_t k = k8 * 8;
    if (!rc.valid || k >= Ktot) return zero;
    uint32_t rs = dKo.div(k), ko = dKo.mod(k, rs);
    uint32_t r = dS.div(rs), s = dS.mod(rs, r);
    int hp = rc.hp - (int)r;
    int wp = rc.wp - (int)s;
    if (hp < 0 || wp < 0) return zero;
    uint32_t p = (uint32_t)hp, q = (uint32_t)wp;
    if (u > 1) { if (hp % u) return zero; p = hp / u; }
    if (v > 1) { if (wp % v) return zero; q = wp / v; }
    if (p >= P || q >= Q) return zero;
    return rc.base_n + ((size_t)p * Q + q) * Ko + ko;
  }
};

// ---------------------------------------------------------------- epilogues

template <typename elem_t>
struct EpiBF16 {
  using elem = elem_t;
  using V8 = typename E8<elem_t>::v8;
  static constexpr bool kLdsStage = true;
  static constexpr bool kBnBwd = false;
  static constexpr bool kFwdStats = false;
  elem_t* c;
  const float* bias;  // nullable
  uint32_t M, N;
  int relu;
  // optional elementwise accumulator: out = gemm + acc (residual-join grad
  // fusion; acc may alias c for in-place accumulate — same thread reads and
  // writes the same element, so aliasing is safe)
  const elem_t* acc = nullptr;
  // coalesced row-chunk store used by the kernel's LDS-staged epilogue.
  // N need not be a multiple of 8 (e.g. FullyConnected num_classes=100):
  // the tail chunk is stored element-wise — a full V8 there would stomp the
  // next row's first columns (and, on the last row, the heap past the
  // tensor), and the bias read would run off the end of the bias vector.
  __device__ __forceinline__ void store_chunk(uint32_t m, uint32_t n0,
                                              V8 v) const {
    if (m >= M || n0 >= N) return;
    const uint32_t rem = N - n0;
    if (acc && rem >= 8) {
      V8 a = *(const V8*)(acc + (size_t)m * N + n0);
#pragma unroll
      for (int e = 0; e < 8; ++e) v[e] = (elem_t)((float)v[e] + (float)a[e]);
    } else if (acc) {
      for (uint32_t e = 0; e < rem; ++e)
        v[e] = (elem_t)((float)v[e] + (float)acc[(size_t)m * N + n0 + e]);
    }
    if (bias || relu) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float f = (float)v[e] + (bias && (uint32_t)e < rem ? bias[n0 + e] : 0.f);
        if (relu) f = fmaxf(f, 0.f);
        v[e] = (elem_t)f;
      }
    }
    if (rem >= 8) {
      *(V8*)(c + (size_t)m * N + n0) = v;
    } else {
      for (uint32_t e = 0; e < rem; ++e) c[(size_t)m * N + n0 + e] = v[e];
    }
  }
  template <int NJ>
  __device__ __forceinline__ void store(const f32x4 (&acc)[4][NJ], uint32_t m0,
                                        uint32_t n0, uint32_t lane) const {
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < NJ; ++j) {
        uint32_t n = n0 + j * 16 + (lane & 15);
        if (n >= N) continue;
        float b = bias ? bias[n] : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          uint32_t m = m0 + i * 16 + ((lane >> 4) << 2) + r;
          if (m >= M) continue;
          float v = acc[i][j][r] + b;
          if (relu) v = fmaxf(v, 0.f);
          c[(size_t)m * N + n] = (elem_t)v;
        }
      }
  }

};

// Epilogue with fused BatchNorm-backward prologue: the
// GEMM computes dy of a BN+ReLU layer (conv dgrad of the consumer conv, or
// the residual-join accumulate). Instead of writing raw dy and re-reading
// (x, dy, y) in a separate bn_bwd_stats pass (HBM-roofline, 10% of the
// ResNet-50 step), the epilogue:
//   - applies the ReLU mask from y in-register (g = y>0 ? dy : 0),
//   - writes the MASKED gradient (downstream BN math never needs y again),
//   - accumulates per-block partial sums db=Σg, dg=Σg*xhat into
//     [tiles_m][N] slabs (no atomics: each block owns its slab row range).
// bn_bwd_finalize_slabs + bn_bwd_dx_presummed complete the BN backward.
// Reference analog: batch_norm.cu:314 BatchNormalizationBackwardKernel,
// restructured as a GEMM-epilogue fusion.
template <typename elem_t>
struct EpiBnBwd {
  using elem = elem_t;
  using V8 = typename E8<elem_t>::v8;
  static constexpr bool kLdsStage = true;
  static constexpr bool kBnBwd = true;
  static constexpr bool kFwdStats = false;
  elem_t* c;             // output: masked gradient g
  const elem_t* acc;     // optional residual-join accumulate (may alias c)
  const elem_t* bnb_y;   // BN(+relu) output (mask source), same [M][N] layout
  const elem_t* bnb_x;   // pre-BN conv output (xhat source)
  const float* bnb_mean;    // [N]
  const float* bnb_invstd;  // [N]
  float* bnb_pdb;           // [tiles_m][N]
  float* bnb_pdg;           // [tiles_m][N]
  uint32_t M, N;
};

// 1x1 stride-u dgrad: rows m = (n,p,q) of the dense dy@W^T GEMM scatter to
// input pixels (n, p*u, q*v); everything else in dx stays zero.
template <typename elem_t>
struct EpiBF16Scatter {
  using elem = elem_t;
  using V8 = typename E8<elem_t>::v8;
  static constexpr bool kLdsStage = true;
  static constexpr bool kBnBwd = false;
  static constexpr bool kFwdStats = false;
  elem_t* dx;
  uint32_t M, N;  // M = NPQ, N = C
  uint32_t H, W, Q;
  int u, v;
  int accumulate = 0;  // dx += val (uncovered pixels keep their value)
  FastDiv dQ, dPQ;
  __device__ __forceinline__ void store_chunk(uint32_t m, uint32_t n0,
                                              V8 val) const {
    if (m >= M || n0 >= N) return;
    uint32_t n = dPQ.div(m), pq = dPQ.mod(m, n);
    uint32_t p = dQ.div(pq), q = dQ.mod(pq, p);
    size_t off = (((size_t)n * H + p * u) * W + q * v) * N + n0;
    if (accumulate) {
      V8 a = *(const V8*)(dx + off);
#pragma unroll
      for (int e = 0; e < 8; ++e) val[e] = (elem_t)((float)val[e] + (float)a[e]);
    }
    *(V8*)(dx + off) = val;
  }
  template <int NJ>
  __device__ __forceinline__ void store(const f32x4 (&acc)[4][NJ], uint32_t,
                                        uint32_t, uint32_t) const {}
};

template <typename elem_t>
struct EpiAtomicF32 {
  using elem = elem_t;
  using V8 = typename E8<elem_t>::v8;  // split-K partial accumulation (conv wgrad)
  static constexpr bool kLdsStage = false;
  static constexpr bool kBnBwd = false;
  static constexpr bool kFwdStats = false;
  float* c;
  uint32_t M, N;
  __device__ __forceinline__ void store_chunk(uint32_t, uint32_t, V8) const {}
  template <int NJ>
  __device__ __forceinline__ void store(const f32x4 (&acc)[4][NJ], uint32_t m0,
                                        uint32_t n0, uint32_t lane) const {
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < NJ; ++j) {
        uint32_t n = n0 + j * 16 + (lane & 15);
        if (n >= N) continue;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          uint32_t m = m0 + i * 16 + ((lane >> 4) << 2) + r;
          if (m >= M) continue;
          atomicAdd(&c[(size_t)m * N + n], acc[i][j][r]);
        }
      }
  }
};

// conv fwd epilogue with fused BN statistics — the SECOND attempt at this
// fusion: round 1 used per-stripe atomicAdd into shared slab rows and
// measured -3.5% end to end; this version reuses the EpiBnBwd scheme
// (fixed 8-channel group per thread + one LDS tree + exclusive slab-row
// writes, no atomics) that measured +6% on the backward side. The slabs are
// column sums/sumsq of the values the epilogue itself writes, so the
// following BatchNorm skips its whole-tensor stats read.
template <typename elem_t>
struct EpiBF16FwdStats {
  using elem = elem_t;
  using V8 = typename E8<elem_t>::v8;
  static constexpr bool kLdsStage = true;
  static constexpr bool kBnBwd = false;
  static constexpr bool kFwdStats = true;
  elem_t* c;
  float* bn_psum;    // [tiles_m][N]
  float* bn_psumsq;  // [tiles_m][N]
  uint32_t M, N;
};

// column-group tree reduction + exclusive slab-row write shared by the
// fused-BN epilogues: threads t ≡ nc (mod NC8) hold partials for channel
// group nc; after the tree, thread t < NC8 writes its 8 channels of the
// block's own slab row. `red` is 8 KiB of post-compute LDS.
template <int NC8, int THREADS = 256>
__device__ __forceinline__ void epi_colreduce_write(
    float* red, const float (&v)[8], float* slab, uint32_t bm, uint32_t bn,
    uint32_t N, uint32_t t) {
  __syncthreads();
#pragma unroll
  for (int e = 0; e < 8; ++e) red[t * 8 + e] = v[e];
  __syncthreads();
  for (uint32_t off = THREADS / 2; off >= NC8; off >>= 1) {
    if (t < off) {
#pragma unroll
      for (int e = 0; e < 8; ++e) red[t * 8 + e] += red[(t + off) * 8 + e];
    }
    __syncthreads();
  }
  if (t < NC8 && bn + t * 8 < N) {
    float* prow = slab + (size_t)(bm >> 7) * N + bn + t * 8;
    uint32_t remw = min(8u, N - (bn + t * 8));
    for (uint32_t e = 0; e < remw; ++e) prow[e] = red[t * 8 + e];
  }
}

// ------------------------------------------------------------------- kernel

template <int NJ, class PA, class PB, class EPI>
__launch_bounds__(256, 2) __global__
void gemm_tn_kernel(PA pa, PB pb, EPI epi, uint32_t ktiles_total,
                    uint32_t tiles_n, uint32_t kt_per_slice) {
  using elem_t = typename PA::elem;
  using V8 = typename E8<elem_t>::v8;
  // NJ = 16-col fragments per wave: BN tile = NJ*32 (128 for square work,
  // 64 for narrow-N layers like resnet's K_out=64 convs where a 128 tile
  // wastes half the MFMA work).
  // split-K: blockIdx.y selects a K-slice (fp32-atomic epilogue makes the
  // slices order-independent); single launch fills the chip.
  constexpr uint32_t BN = NJ * 32;
  const uint32_t kt0 = blockIdx.y * kt_per_slice;
  const uint32_t ktiles = min(kt_per_slice, ktiles_total - kt0);
  if (kt0 >= ktiles_total) return;
  __shared__ elem_t smem[2][(128 + BN) * 64];  // A then B images per buffer
  constexpr uint32_t B_OFF = 128 * 64;
  const uint32_t t = threadIdx.x;
  const uint32_t wave = t >> 6, lane = t & 63;
  const uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const uint32_t bm = (bid / tiles_n) * 128, bn = (bid % tiles_n) * BN;

  // staging geometry: iteration it covers rows it*32+(t>>3), byte col (t&7)*16;
  // XOR swizzle applied to the SOURCE k-piece so the LDS image stays
  // lane-linear for global_load_lds (rule 21).
  uint32_t srow[4], sk8[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    srow[it] = it * 32 + (t >> 3);
    uint32_t colb = (t & 7) * 16;
    sk8[it] = (colb ^ ((srow[it] & 7) << 4)) >> 4;
  }
  // per-block row contexts: the expensive (n,p,q)/(n,h,w) decode runs once
  typename PA::Row arow[4];
  typename PB::Row brow[NJ];
#pragma unroll
  for (int it = 0; it < 4; ++it) arow[it] = pa.row(bm + srow[it]);
#pragma unroll
  for (int it = 0; it < NJ; ++it) brow[it] = pb.row(bn + srow[it]);

  auto stage = [&](int buf, uint32_t kt) {
#pragma unroll
    for (int it = 0; it < 4; ++it)
      glds16(pa.addr(arow[it], kt * 8 + sk8[it]),
             &smem[buf][it * 2048 + wave * 512]);
#pragma unroll
    for (int it = 0; it < NJ; ++it)
      glds16(pb.addr(brow[it], kt * 8 + sk8[it]),
             &smem[buf][B_OFF + it * 2048 + wave * 512]);
  };

  const uint32_t wr = (wave >> 1) * 64, wc = (wave & 1) * NJ * 16;
  f32x4 acc[4][NJ] = {};

  // 2-deep glds pipeline with COUNTED vmcnt + raw barriers (cdna_hip_
  // programming.md §5 'Pipelining across barriers'): tile t+1's LDS-DMA
  // stays in flight across the barrier while tile t computes — a vmcnt(0)
  // drain per tile exposes the full HBM round trip at every K-step
  // (measured ~4x off on resnet's short-K conv shapes).
  constexpr int G = 4 + NJ;  // glds issued per wave per tile
  auto wait_tile = [&](bool one_in_flight) {
    if (one_in_flight) {
      if constexpr (NJ == 4)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
  };

  stage(0, kt0);
  if (ktiles > 1) stage(1, kt0 + 1);
  for (uint32_t kt = 0; kt < ktiles; ++kt) {
    const uint32_t cur = kt & 1;
    wait_tile(kt + 1 < ktiles);          // tile kt landed (kt+1 may fly on)
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      V8 af[4], bfr[NJ];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        uint32_t row = wr + i * 16 + (lane & 15);
        uint32_t colb = (kk * 64 + ((lane >> 4) << 4)) ^ ((row & 7) << 4);
        af[i] = *(const V8*)((const char*)&smem[cur][0] + row * 128 + colb);
      }
#pragma unroll
      for (int j = 0; j < NJ; ++j) {
        uint32_t row = wc + j * 16 + (lane & 15);
        uint32_t colb = (kk * 64 + ((lane >> 4) << 4)) ^ ((row & 7) << 4);
        bfr[j] = *(const V8*)((const char*)&smem[cur][B_OFF] + row * 128 + colb);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < NJ; ++j)
          acc[i][j] = E8<elem_t>::mfma(af[i], bfr[j], acc[i][j]);
      __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_s_barrier();        // every wave done reading smem[cur]
    if (kt + 2 < ktiles) stage(cur, kt0 + kt + 2);
  }
  if constexpr (EPI::kLdsStage) {
    // stage the C tile through LDS so global stores are 16-B row chunks:
    // the direct fragment store is 16-64 half-coalesced 2-B stores per lane
    // (store-issue-bound; cdna_hip_programming.md T21 diagnosis).
    __syncthreads();  // nothing in flight; reuse smem[0] as [128][BN] bf16
    elem_t* ct = &smem[0][0];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < NJ; ++j) {
        uint32_t col = wc + j * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          uint32_t row = wr + i * 16 + ((lane >> 4) << 2) + r;
          ct[row * BN + col] = (elem_t)acc[i][j][r];
        }
      }
    __syncthreads();
    constexpr uint32_t CHUNKS = 128 * BN / 8;
    if constexpr (EPI::kBnBwd) {
      // BN-backward fused epilogue: mask with relu'(y), write masked g,
      // accumulate db=Σg, dg=Σg*(x-mean)*invstd per channel. Chunk stride
      // 256 with BN/8 | 256 keeps each thread on a FIXED 8-channel group,
      // so the sums live in registers and one LDS tree finishes the block.
      constexpr uint32_t NC8 = BN / 8;
      const uint32_t nc = t % NC8;
      const uint32_t n0 = bn + nc * 8;
      float mean8[8], inv8[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        uint32_t n = n0 + e;
        mean8[e] = n < epi.N ? epi.bnb_mean[n] : 0.f;
        inv8[e] = n < epi.N ? epi.bnb_invstd[n] : 0.f;
      }
      float db[8] = {}, dg[8] = {};
      for (uint32_t idx = t; idx < CHUNKS; idx += 256) {
        uint32_t row = idx / NC8;
        uint32_t m = bm + row;
        if (m >= epi.M || n0 >= epi.N) continue;
        V8 v = *(const V8*)(ct + row * BN + nc * 8);
        size_t off = (size_t)m * epi.N + n0;
        uint32_t rem = epi.N - n0;
        if (rem >= 8) {
          V8 a{};
          if (epi.acc) a = *(const V8*)(epi.acc + off);
          V8 yv = *(const V8*)(epi.bnb_y + off);
          V8 xv = *(const V8*)(epi.bnb_x + off);
          V8 g;
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            float f = (float)v[e] + (epi.acc ? (float)a[e] : 0.f);
            if ((float)yv[e] <= 0.f) f = 0.f;
            g[e] = (elem_t)f;
            db[e] += f;
            dg[e] += f * ((float)xv[e] - mean8[e]) * inv8[e];
          }
          *(V8*)(epi.c + off) = g;
        } else {
          for (uint32_t e = 0; e < rem; ++e) {
            float f = (float)v[e] + (epi.acc ? (float)epi.acc[off + e] : 0.f);
            if ((float)epi.bnb_y[off + e] <= 0.f) f = 0.f;
            epi.c[off + e] = (elem_t)f;
            db[e] += f;
            dg[e] += f * ((float)epi.bnb_x[off + e] - mean8[e]) * inv8[e];
          }
        }
      }
      // tree-reduce + exclusive slab-row writes — no atomics (vs round 1's
      // measured atomic cost); smem[1] is drained, reuse as 8 KiB scratch
      float* red = (float*)&smem[1][0];
      epi_colreduce_write<(int)NC8>(red, db, epi.bnb_pdb, bm, bn, epi.N, t);
      epi_colreduce_write<(int)NC8>(red, dg, epi.bnb_pdg, bm, bn, epi.N, t);
    } else if constexpr (EPI::kFwdStats) {
      // fused forward BN statistics: column sums/sumsq of the tile the
      // epilogue writes — the following BatchNorm skips its stats pass
      constexpr uint32_t NC8 = BN / 8;
      const uint32_t nc = t % NC8;
      const uint32_t n0 = bn + nc * 8;
      float s[8] = {}, ss[8] = {};
      for (uint32_t idx = t; idx < CHUNKS; idx += 256) {
        uint32_t row = idx / NC8;
        uint32_t m = bm + row;
        if (m >= epi.M || n0 >= epi.N) continue;
        V8 v = *(const V8*)(ct + row * BN + nc * 8);
        size_t off = (size_t)m * epi.N + n0;
        uint32_t rem = epi.N - n0;
        if (rem >= 8) {
          *(V8*)(epi.c + off) = v;
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            float f = (float)v[e];
            s[e] += f;
            ss[e] += f * f;
          }
        } else {
          for (uint32_t e = 0; e < rem; ++e) {
            epi.c[off + e] = v[e];
            float f = (float)v[e];
            s[e] += f;
            ss[e] += f * f;
          }
        }
      }
      float* red = (float*)&smem[1][0];
      epi_colreduce_write<(int)NC8>(red, s, epi.bn_psum, bm, bn, epi.N, t);
      epi_colreduce_write<(int)NC8>(red, ss, epi.bn_psumsq, bm, bn, epi.N, t);
    } else {
      for (uint32_t idx = t; idx < CHUNKS; idx += 256) {
        uint32_t row = idx / (BN / 8), nc = idx % (BN / 8);
        epi.store_chunk(bm + row, bn + nc * 8,
                        *(const V8*)(ct + row * BN + nc * 8));
      }
    }
  } else {
    epi.template store<NJ>(acc, bm + wr, bn + wc, lane);
  }
}

// ------------------------------------------ 256^2 8-phase dense TN kernel
// The deep-pipelined big-tile path (cdna_hip_programming.md §5 "256² 8-phase
// template"; standalone probe: gemm256.hip — 1200 TF @8192³ vs 892 for the
// 128² structure). Dense providers only (DenseP both sides); carries the
// full fused-epilogue family (EpiBF16 / EpiBnBwd / EpiBF16FwdStats) so the
// production 1x1-conv and FC GEMMs can route here. Schedule: per K-tile,
// phase 0 reads ALL B fragments to registers (+ first A quarter), phases
// 1-3 one A quarter each; one half-tile of the {B0,B1,A0,A1} stream staged
// per phase with a 7-half prologue and vmcnt(6) at tile boundaries only.
template <class PA, class PB, class EPI>
__launch_bounds__(512, 2) __global__
void gemm256f_kernel(PA pa, PB pb, EPI epi, uint32_t ktiles, uint32_t tiles_n) {
  using elem_t = typename PA::elem;
  using V8 = typename E8<elem_t>::v8;
  __shared__ __attribute__((aligned(16))) elem_t smem[2][4][8192];
  const uint32_t t = threadIdx.x;
  const uint32_t wave = t >> 6, lane = t & 63;
  const uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  const uint32_t bm = (bid / tiles_n) * 256, bn = (bid % tiles_n) * 256;
  const uint32_t am_half = wave >> 2;
  const uint32_t wc = (wave & 3) * 64;

  // row-parity source swizzle of the 128² kernel (beat st_16x32 by ~2% with
  // 0.2% vs 2.5% bank-conflict cycles on this layout)
  auto swz = [](uint32_t b) { return b ^ (((b >> 7) & 7) << 4); };

  // staging sources. Dense sides get branch-free precomputed pointers with
  // a per-tile stride (0 for OOB rows -> zero page; K%64==0 enforced by the
  // caller). Gather sides (conv im2col/dgrad A) hoist the per-row decode
  // into a Row context once (the same trick as the 128^2 kernel) and call
  // addr(ctx, k8) per staged piece.
  const elem_t* src0[4][2];
  size_t sstep[4][2];
  typename PA::Row actx[2][2];
  typename PB::Row bctx[2][2];
  uint32_t k8l[4][2];
#pragma unroll
  for (uint32_t part = 0; part < 4; ++part)
#pragma unroll
    for (uint32_t g = 0; g < 2; ++g) {
      uint32_t off = (g * 8 + wave) * 1024 + lane * 16;
      uint32_t lb = swz(off);
      uint32_t row = lb >> 7, kb = lb & 127;
      uint32_t k = (kb >> 4) * 8;
      k8l[part][g] = kb >> 4;
      if (part < 2) {
        uint32_t m = bn + part * 128 + row;
        if constexpr (PB::kDense) {
          bool oob = m >= pb.M;
          src0[part][g] = oob ? pb.zero : pb.base + (size_t)m * pb.ld + k;
          sstep[part][g] = oob ? 0 : 64;
        } else {
          bctx[part][g] = pb.row(m);
        }
      } else {
        uint32_t m = bm + (part - 2) * 128 + row;
        if constexpr (PA::kDense) {
          bool oob = m >= pa.M;
          src0[part][g] = oob ? pa.zero : pa.base + (size_t)m * pa.ld + k;
          sstep[part][g] = oob ? 0 : 64;
        } else {
          actx[part - 2][g] = pa.row(m);
        }
      }
    }
  auto stage_part = [&](auto part_c, uint32_t buf, uint32_t kt) {
    constexpr uint32_t part = decltype(part_c)::value;
#pragma unroll
    for (uint32_t g = 0; g < 2; ++g) {
      void* dst = &smem[buf][part][(g * 8 + wave) * 512];
      if constexpr (part < 2) {
        if constexpr (PB::kDense)
          glds16(src0[part][g] + (size_t)kt * sstep[part][g], dst);
        else
          glds16(pb.addr(bctx[part][g], kt * 8 + k8l[part][g]), dst);
      } else {
        if constexpr (PA::kDense)
          glds16(src0[part][g] + (size_t)kt * sstep[part][g], dst);
        else
          glds16(pa.addr(actx[part - 2][g], kt * 8 + k8l[part][g]), dst);
      }
    }
  };
  const uint32_t total_halves = ktiles * 4;
  auto stage_stream = [&](uint32_t h) {
    if (h >= total_halves) return;
    uint32_t kt = h >> 2, buf = kt & 1;
    switch (h & 3) {
      case 0: stage_part(std::integral_constant<uint32_t, 0>{}, buf, kt); break;
      case 1: stage_part(std::integral_constant<uint32_t, 1>{}, buf, kt); break;
      case 2: stage_part(std::integral_constant<uint32_t, 2>{}, buf, kt); break;
      default: stage_part(std::integral_constant<uint32_t, 3>{}, buf, kt);
    }
  };
  auto wait_vm = [&](uint32_t n) {
    if (n >= 6)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else if (n == 4)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if (n == 2)
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  };

  uint32_t issued = 0;
  for (; issued < min(4u, total_halves); ++issued) stage_stream(issued);
  wait_vm(total_halves > 4 ? 4 : 0);
  for (; issued < min(7u, total_halves); ++issued) stage_stream(issued);
  wait_vm(2 * (issued - min(4u, total_halves)));
  __builtin_amdgcn_s_barrier();

  f32x4 acc[8][4] = {};
  const uint32_t a_slot = 2 + am_half;
  const uint32_t b_slot = wc >> 7;
  const uint32_t wc_local = wc & 127;

  V8 bf[8];
  uint32_t P = 0;
  for (uint32_t kt = 0; kt < ktiles; ++kt) {
    const uint32_t cur = kt & 1;
#pragma unroll
    for (uint32_t q = 0; q < 4; ++q, ++P) {
      if (q == 0) {
#pragma unroll
        for (uint32_t j = 0; j < 4; ++j)
#pragma unroll
          for (uint32_t kk = 0; kk < 2; ++kk) {
            uint32_t row = wc_local + j * 16 + (lane & 15);
            uint32_t kbyte = kk * 64 + ((lane >> 4) << 4);
            bf[j * 2 + kk] = *(const V8*)((const char*)&smem[cur][b_slot][0] +
                                          swz(row * 128 + kbyte));
          }
      }
      V8 af[2][2];
#pragma unroll
      for (uint32_t ii = 0; ii < 2; ++ii)
#pragma unroll
        for (uint32_t kk = 0; kk < 2; ++kk) {
          uint32_t row = (2 * q + ii) * 16 + (lane & 15);
          uint32_t kbyte = kk * 64 + ((lane >> 4) << 4);
          af[ii][kk] = *(const V8*)((const char*)&smem[cur][a_slot][0] +
                                    swz(row * 128 + kbyte));
        }
      stage_stream(7 + P);
      if (q == 0)
        asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (uint32_t kk = 0; kk < 2; ++kk)
#pragma unroll
        for (uint32_t ii = 0; ii < 2; ++ii)
#pragma unroll
          for (uint32_t j = 0; j < 4; ++j)
            acc[2 * q + ii][j] = E8<elem_t>::mfma(af[ii][kk], bf[j * 2 + kk],
                                                  acc[2 * q + ii][j]);
      __builtin_amdgcn_s_setprio(0);
      if (q == 3) {
        uint32_t issued_now = min(total_halves, 7 + P + 1);
        uint32_t needed = min(total_halves, 4 * (kt + 2));
        wait_vm(2 * (issued_now - needed));
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  // LDS-staged epilogue in two 128-row half passes (the C tile is 128 KiB =
  // the whole LDS; ct of one half = 64 KiB). Same chunk/tree machinery as
  // the 128² kernel, scaled to 512 threads / 32 chunks per row.
  elem_t* ct = &smem[0][0][0];  // [128 rows][256 cols] per pass
  constexpr uint32_t NC8 = 32;
  const uint32_t nc = t % NC8;
  const uint32_t n0 = bn + nc * 8;
#pragma unroll
  for (uint32_t h = 0; h < 2; ++h) {
    __syncthreads();  // no glds in flight after the final boundary wait
    if (am_half == h) {
#pragma unroll
      for (uint32_t i = 0; i < 8; ++i)
#pragma unroll
        for (uint32_t j = 0; j < 4; ++j) {
          uint32_t col = wc + j * 16 + (lane & 15);
#pragma unroll
          for (uint32_t r = 0; r < 4; ++r) {
            uint32_t row = i * 16 + ((lane >> 4) << 2) + r;
            ct[row * 256 + col] = (elem_t)acc[i][j][r];
          }
        }
    }
    __syncthreads();
    const uint32_t bmh = bm + h * 128;
    constexpr uint32_t CHUNKS = 128 * NC8;
    if constexpr (EPI::kBnBwd) {
      float mean8[8], inv8[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        uint32_t n = n0 + e;
        mean8[e] = n < epi.N ? epi.bnb_mean[n] : 0.f;
        inv8[e] = n < epi.N ? epi.bnb_invstd[n] : 0.f;
      }
      float db[8] = {}, dg[8] = {};
      for (uint32_t idx = t; idx < CHUNKS; idx += 512) {
        uint32_t row = idx / NC8;
        uint32_t m = bmh + row;
        if (m >= epi.M || n0 >= epi.N) continue;
        V8 v = *(const V8*)(ct + row * 256 + nc * 8);
        size_t off = (size_t)m * epi.N + n0;
        uint32_t rem = epi.N - n0;
        if (rem >= 8) {
          V8 a{};
          if (epi.acc) a = *(const V8*)(epi.acc + off);
          V8 yv = *(const V8*)(epi.bnb_y + off);
          V8 xv = *(const V8*)(epi.bnb_x + off);
          V8 g;
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            float f = (float)v[e] + (epi.acc ? (float)a[e] : 0.f);
            if ((float)yv[e] <= 0.f) f = 0.f;
            g[e] = (elem_t)f;
            db[e] += f;
            dg[e] += f * ((float)xv[e] - mean8[e]) * inv8[e];
          }
          *(V8*)(epi.c + off) = g;
        } else {
          for (uint32_t e = 0; e < rem; ++e) {
            float f = (float)v[e] + (epi.acc ? (float)epi.acc[off + e] : 0.f);
            if ((float)epi.bnb_y[off + e] <= 0.f) f = 0.f;
            epi.c[off + e] = (elem_t)f;
            db[e] += f;
            dg[e] += f * ((float)epi.bnb_x[off + e] - mean8[e]) * inv8[e];
          }
        }
      }
      float* red = (float*)ct;  // chunk reads done after the next barrier
      epi_colreduce_write<NC8, 512>(red, db, epi.bnb_pdb, bmh, bn, epi.N, t);
      epi_colreduce_write<NC8, 512>(red, dg, epi.bnb_pdg, bmh, bn, epi.N, t);
    } else if constexpr (EPI::kFwdStats) {
      float s[8] = {}, ss[8] = {};
      for (uint32_t idx = t; idx < CHUNKS; idx += 512) {
        uint32_t row = idx / NC8;
        uint32_t m = bmh + row;
        if (m >= epi.M || n0 >= epi.N) continue;
        V8 v = *(const V8*)(ct + row * 256 + nc * 8);
        size_t off = (size_t)m * epi.N + n0;
        uint32_t rem = epi.N - n0;
        if (rem >= 8) {
          *(V8*)(epi.c + off) = v;
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            float f = (float)v[e];
            s[e] += f;
            ss[e] += f * f;
          }
        } else {
          for (uint32_t e = 0; e < rem; ++e) {
            epi.c[off + e] = v[e];
            float f = (float)v[e];
            s[e] += f;
            ss[e] += f * f;
          }
        }
      }
      float* red = (float*)ct;
      epi_colreduce_write<NC8, 512>(red, s, epi.bn_psum, bmh, bn, epi.N, t);
      epi_colreduce_write<NC8, 512>(red, ss, epi.bn_psumsq, bmh, bn, epi.N, t);
    } else {
      for (uint32_t idx = t; idx < CHUNKS; idx += 512) {
        uint32_t row = idx / NC8;
        epi.store_chunk(bmh + row, bn + (idx % NC8) * 8,
                        *(const V8*)(ct + row * 256 + (idx % NC8) * 8));
      }
    }
  }
}

// -------------------------------------------- NT kernel (contraction-major)
// Both operands stored [Kd][cols] (contraction-major) — conv wgrad consumes
// dy and the im2col view of x NHWC-NATIVE, eliminating the dy^T/im2col^T
// materialization passes (they were ~2x the wgrad GEMM's own HBM traffic).
// LDS images [64 kd][BM|BN]; fragments gathered by scalar u16 column reads
// (8 per fragment), hidden under the MFMA stream.

template <typename elem_t>
struct WgradDyA {
  using elem = elem_t;  // A: [kd = npq][m = ko] = dy NHWC as-is
  const elem_t* dy;
  const elem_t* zero;
  uint32_t Kd, Mdim;
  __device__ __forceinline__ const void* addr(uint32_t kd, uint32_t c8) const {
    if (kd >= Kd || c8 * 8 >= Mdim) return zero;
    return dy + (size_t)kd * Mdim + c8 * 8;
  }
};

template <typename elem_t>
struct WgradXcolB {
  using elem = elem_t;  // B: [kd = npq][n = (r,s,c)] gathered from x NHWC
  const elem_t* x;
  const elem_t* zero;
  uint32_t Kd, Ndim;  // Kd = NPQ, Ndim = R*S*C
  uint32_t C, H, W, Q, S;
  int u, v, ph, pw;
  FastDiv dQ, dPQ, dC, dS;
  __device__ __forceinline__ const void* addr(uint32_t kd, uint32_t c8) const {
    uint32_t nn = c8 * 8;
    if (kd >= Kd || nn >= Ndim) return zero;
    uint32_t n = dPQ.div(kd), pq = dPQ.mod(kd, n);
    uint32_t p = dQ.div(pq), q = dQ.mod(pq, p);
    uint32_t rs = dC.div(nn), c = dC.mod(nn, rs);
    uint32_t r = dS.div(rs), s = dS.mod(rs, r);
    int ih = (int)(p * u) - ph + (int)r;
    int iw = (int)(q * v) - pw + (int)s;
    if ((uint32_t)ih >= H || (uint32_t)iw >= W) return zero;
    return x + (((size_t)n * H + ih) * W + iw) * C + c;
  }
};

// WM = waves along M (wave grid WM x (4/WM)): WM=2 is the square 128xBN
// tile; WM=1 is the flat 64x(NJ*64) tile for small-M work — conv wgrad's
// M = K_out is 64 on ResNet's narrow layers, where the 128-row tile wastes
// HALF its MFMA work on zero-padded rows (measured 222 TF on l1.conv2).
// DEPTH: LDS K-tile ring depth. 2 = classic double buffer (ONE tile in
// flight across the compute of the next — the ~550-cycle MFMA body of a
// tile does not cover the ~900-cycle HBM round trip, measured 35% parked
// cycles on wgrad). 3 = two tiles in flight (costs a third of the LDS;
// occupancy drops to 1 block/CU at NJ=4 — A/B via DTMX_NT_DEPTH).
// GROUP: XCD-grouped 1-D launch (DTMX_NT_GROUP): all output tiles of one
// split-K slice are mapped onto ONE XCD (hardware round-robins 1-D block
// slots across the 8 XCDs, so slot%8 selects the die). The tiles of a
// slice stream the SAME 16-32 KiB dy K-tile per step — co-locating them
// turns the tiles_n-fold dy re-read (18x on a 256-channel 3x3 wgrad, the
// dominant HBM traffic of the whole wgrad family) into XCD-L2 hits.
template <int NJ, int WM, int DEPTH, class PA, class PB, class EPI,
          int GROUP = 0>
__launch_bounds__(256, 2) __global__
void gemm_nt_kernel(PA pa, PB pb, EPI epi, uint32_t ktiles_total,
                    uint32_t tiles_n, uint32_t kt_per_slice,
                    uint32_t tiles_total = 0) {
  using elem_t = typename PA::elem;
  using V8 = typename E8<elem_t>::v8;
  constexpr uint32_t WN = 4 / WM;
  constexpr uint32_t BM = WM * 64;
  constexpr uint32_t BN = WN * NJ * 16;
  uint32_t slice;
  uint32_t bid_raw;
  if constexpr (GROUP) {
    const uint32_t xcd = blockIdx.x & 7, j = blockIdx.x >> 3;
    bid_raw = j % tiles_total;
    slice = xcd + 8 * (j / tiles_total);
  } else {
    bid_raw = blockIdx.x;
    slice = blockIdx.y;
  }
  const uint32_t kt0 = slice * kt_per_slice;
  const uint32_t ktiles = min(kt_per_slice, ktiles_total - kt0);
  if (kt0 >= ktiles_total) return;
  // 16-B alignment: ds_read_b64_tr_b16 at a misaligned address silently
  // returns the 8-aligned address's data (G17)
  __shared__ __attribute__((aligned(16))) elem_t smem[DEPTH][64 * (BM + BN)];
  constexpr uint32_t B_OFF = 64 * BM;
  const uint32_t t = threadIdx.x;
  const uint32_t wave = t >> 6, lane = t & 63;
  const uint32_t bid = GROUP ? bid_raw : xcd_swizzle(bid_raw, gridDim.x);
  const uint32_t bm = (bid / tiles_n) * BM, bn = (bid % tiles_n) * BN;

  // staging: A rows are 256 B (128 m), chunk = 16 kd-rows per 4 KiB;
  //          B rows are BN*2 B, NJ 4-KiB chunks total.
  // column-XOR swizzle by kd-row bit 3: the two 16-lane clusters of a
  // 32-lane read half sit at kd and kd+8, which land on identical banks in
  // a linear image (2-way conflict, SQ_LDS_BANK_CONFLICT ~3e9/step measured)
  // — flipping 16 columns on odd (kd>>3) separates them. Applied on the
  // glds SOURCE (lane-linear dest, rule 21) and un-applied on the reads.
  // ITS_A/ITS_B glds instructions per tile: each covers 256 lanes x 16 B
  constexpr uint32_t PPR_A = BM / 8;   // 16-B pieces per kd-row of A
  constexpr uint32_t PPR_B = BN / 8;
  constexpr uint32_t ITS_A = 64 * PPR_A / 256;
  constexpr uint32_t ITS_B = 64 * PPR_B / 256;
  auto stage = [&](int buf, uint32_t kt) {
    const uint32_t kd0 = kt * 64;
#pragma unroll
    for (uint32_t it = 0; it < ITS_A; ++it) {
      uint32_t krow = it * (256 / PPR_A) + t / PPR_A;
      uint32_t piece = (t % PPR_A) ^ (((krow >> 3) & 1) << 1);
      glds16(pa.addr(kd0 + krow, bm / 8 + piece),
             &smem[buf][it * 2048 + wave * 512]);
    }
#pragma unroll
    for (uint32_t it = 0; it < ITS_B; ++it) {
      uint32_t krow = it * (256 / PPR_B) + t / PPR_B;
      uint32_t piece = (t % PPR_B) ^ (((krow >> 3) & 1) << 1);
      glds16(pb.addr(kd0 + krow, bn / 8 + piece),
             &smem[buf][B_OFF + it * 2048 + wave * 512]);
    }
  };

  const uint32_t wr = (wave / WN) * 64, wc = (wave % WN) * NJ * 16;
  f32x4 acc[4][NJ] = {};

  constexpr int G = (int)(ITS_A + ITS_B);
  auto wait_tile = [&](uint32_t flight) {
    if (DEPTH == 3 && flight >= 2) {
      if constexpr (G == 10)
        asm volatile("s_waitcnt vmcnt(20)" ::: "memory");
      else if constexpr (G == 8)
        asm volatile("s_waitcnt vmcnt(16)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
    } else if (flight >= 1) {
      if constexpr (G == 10)
        asm volatile("s_waitcnt vmcnt(10)" ::: "memory");
      else if constexpr (G == 8)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
  };

  for (uint32_t d = 0; d < (uint32_t)DEPTH && d < ktiles; ++d)
    stage(d, kt0 + d);
  for (uint32_t kt = 0; kt < ktiles; ++kt) {
    const uint32_t cur = kt % DEPTH;
    wait_tile(min((uint32_t)(DEPTH - 1), ktiles - 1 - kt));
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      V8 af[4], bfr[NJ];
      // hardware transpose reads: each 16-lane group reads a [4 kd][16 col]
      // block of the contraction-major image and receives it column-wise —
      // 2 ds_read_b64_tr_b16 per fragment instead of 8 scalar u16 reads
      // (cdna_hip_programming.md T10; source lane r supplies the 8-B run at
      // row kbase+r/4, cols colbase+(r%4)*4).
      const uint32_t r_ = lane & 15;
      const uint32_t kq = r_ >> 2, cq = (r_ & 3) * 4;
      const uint32_t kg = kk * 32 + ((lane >> 4) << 3) + kq;
      typedef __attribute__((ext_vector_type(2))) unsigned int u32x2_;
      union Frag {
        V8 f;
        u32x2_ h[2];
      };
      {
        unsigned a0, a1, a2, a3, a4, a5, a6, a7;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          uint32_t lo = (kg)*BM + ((wr + i * 16 + cq) ^ (((kg >> 3) & 1) << 4));
          uint32_t hi = (kg + 4) * BM +
                        ((wr + i * 16 + cq) ^ ((((kg + 4) >> 3) & 1) << 4));
          unsigned alo = (unsigned)(uintptr_t)&smem[cur][lo];
          unsigned ahi = (unsigned)(uintptr_t)&smem[cur][hi];
          if (i == 0) { a0 = alo; a1 = ahi; }
          if (i == 1) { a2 = alo; a3 = ahi; }
          if (i == 2) { a4 = alo; a5 = ahi; }
          if (i == 3) { a6 = alo; a7 = ahi; }
        }
        u32x2_ t0, t1, t2, t3, t4, t5, t6, t7;
        asm volatile(
            "ds_read_b64_tr_b16 %0, %8\n\t"
            "ds_read_b64_tr_b16 %1, %9\n\t"
            "ds_read_b64_tr_b16 %2, %10\n\t"
            "ds_read_b64_tr_b16 %3, %11\n\t"
            "ds_read_b64_tr_b16 %4, %12\n\t"
            "ds_read_b64_tr_b16 %5, %13\n\t"
            "ds_read_b64_tr_b16 %6, %14\n\t"
            "ds_read_b64_tr_b16 %7, %15\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=&v"(t0), "=&v"(t1), "=&v"(t2), "=&v"(t3), "=&v"(t4), "=&v"(t5),
              "=&v"(t6), "=&v"(t7)
            : "v"(a0), "v"(a1), "v"(a2), "v"(a3), "v"(a4), "v"(a5), "v"(a6),
              "v"(a7)
            : "memory");
        __builtin_amdgcn_sched_barrier(0);  // rule 18: pin MFMA below the
        // inline-asm lgkmcnt (hipcc may hoist register-only MFMA past it)
        Frag f;
        f.h[0] = t0; f.h[1] = t1; af[0] = f.f;
        f.h[0] = t2; f.h[1] = t3; af[1] = f.f;
        f.h[0] = t4; f.h[1] = t5; af[2] = f.f;
        f.h[0] = t6; f.h[1] = t7; af[3] = f.f;
      }
      {
        unsigned b0, b1, b2, b3, b4, b5, b6, b7;
#pragma unroll
        for (int j = 0; j < NJ; ++j) {
          uint32_t lo = B_OFF + (kg)*BN + ((wc + j * 16 + cq) ^ (((kg >> 3) & 1) << 4));
          uint32_t hi = B_OFF + (kg + 4) * BN +
                        ((wc + j * 16 + cq) ^ ((((kg + 4) >> 3) & 1) << 4));
          unsigned alo = (unsigned)(uintptr_t)&smem[cur][lo];
          unsigned ahi = (unsigned)(uintptr_t)&smem[cur][hi];
          if (j == 0) { b0 = alo; b1 = ahi; }
          if (j == 1) { b2 = alo; b3 = ahi; }
          if (NJ > 2 && j == 2) { b4 = alo; b5 = ahi; }
          if (NJ > 2 && j == 3) { b6 = alo; b7 = ahi; }
        }
        u32x2_ t0, t1, t2, t3;
        asm volatile(
            "ds_read_b64_tr_b16 %0, %4\n\t"
            "ds_read_b64_tr_b16 %1, %5\n\t"
            "ds_read_b64_tr_b16 %2, %6\n\t"
            "ds_read_b64_tr_b16 %3, %7\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=&v"(t0), "=&v"(t1), "=&v"(t2), "=&v"(t3)
            : "v"(b0), "v"(b1), "v"(b2), "v"(b3)
            : "memory");
        __builtin_amdgcn_sched_barrier(0);
        Frag f;
        f.h[0] = t0; f.h[1] = t1; bfr[0] = f.f;
        f.h[0] = t2; f.h[1] = t3; bfr[1] = f.f;
        if (NJ > 2) {
          u32x2_ t4, t5, t6, t7;
          asm volatile(
              "ds_read_b64_tr_b16 %0, %4\n\t"
              "ds_read_b64_tr_b16 %1, %5\n\t"
              "ds_read_b64_tr_b16 %2, %6\n\t"
              "ds_read_b64_tr_b16 %3, %7\n\t"
              "s_waitcnt lgkmcnt(0)"
              : "=&v"(t4), "=&v"(t5), "=&v"(t6), "=&v"(t7)
              : "v"(b4), "v"(b5), "v"(b6), "v"(b7)
              : "memory");
          __builtin_amdgcn_sched_barrier(0);
          f.h[0] = t4; f.h[1] = t5; bfr[2] = f.f;
          f.h[0] = t6; f.h[1] = t7; bfr[3] = f.f;
        }
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < NJ; ++j)
          acc[i][j] = E8<elem_t>::mfma(af[i], bfr[j], acc[i][j]);
      __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_s_barrier();
    if (kt + DEPTH < ktiles) stage(cur, kt0 + kt + DEPTH);
  }
  epi.template store<NJ>(acc, bm + wr, bn + wc, lane);
}

// --------------------------------------------------- transpose (row-gather)
// out[c][m] = row(m)[c]  for generic [M][C] -> [C][out_ld] bf16 transposes
// (linear dgrad/wgrad operands, conv-wgrad dy^T and gathered im2col^T).
// C % 8 == 0 required; rows beyond M (or gather-invalid) write zeros.

template <typename elem_t>
struct IdentityRows {
  using elem = elem_t;
  const elem_t* base;
  uint32_t M, ld;
  __device__ __forceinline__ const elem_t* row(uint32_t m) const {
    return m < M ? base + (size_t)m * ld : nullptr;
  }
};

// rows of the im2col matrix for a FIXED (r,s): row m = (n,p,q) -> x pixel
template <typename elem_t>
struct Im2colRows {
  using elem = elem_t;
  const elem_t* x;
  uint32_t M, C, H, W, Q;
  int u, v, ph, pw, r, s;
  FastDiv dQ, dPQ;
  __device__ __forceinline__ const elem_t* row(uint32_t m) const {
    if (m >= M) return nullptr;
    uint32_t n = dPQ.div(m), pq = dPQ.mod(m, n);
    uint32_t p = dQ.div(pq), q = dQ.mod(pq, p);
    int ih = (int)(p * u) - ph + r;
    int iw = (int)(q * v) - pw + s;
    if ((uint32_t)ih >= H || (uint32_t)iw >= W) return nullptr;
    return x + (((size_t)n * H + ih) * W + iw) * C;
  }
};

template <class ROWS>
__global__ void transpose_rowgather_kernel(ROWS rows, typename ROWS::elem* out,
                                           uint32_t C, uint32_t out_ld,
                                           uint32_t row_off, uint32_t tiles_m) {
  using elem_t = typename ROWS::elem;
  using V8 = typename E8<elem_t>::v8;
  __shared__ elem_t tile[64][72];  // +8 pad (16 B) against bank conflicts
  const uint32_t t = threadIdx.x;
  const uint32_t bm = (blockIdx.x % tiles_m) * 64;   // input row tile
  const uint32_t bc = (blockIdx.x / tiles_m) * 64;   // input col tile
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    uint32_t rl = it * 32 + (t >> 3), cl = (t & 7) * 8;
    const elem_t* p = rows.row(bm + rl);
    V8 v = {};
    if (p && bc + cl < C) v = *(const V8*)(p + bc + cl);
    *(V8*)&tile[rl][cl] = v;
  }
  __syncthreads();
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    uint32_t cl = it * 32 + (t >> 3), ml = (t & 7) * 8;
    if (bc + cl >= C || bm + ml >= out_ld) continue;
    V8 v;
#pragma unroll
    for (int e = 0; e < 8; ++e) v[e] = tile[ml + e][cl];
    *(V8*)&out[(size_t)(row_off + bc + cl) * out_ld + bm + ml] = v;
  }
}

// ------------------------------------------------------- im2col (small-C)
// Explicit im2col for C % 8 != 0 (the 3-channel stem): out[m][k] with
// k = (r*S+s)*C + c, zero-padded to Kpad (reference nn/im2col.cuh analog).
template <typename elem_t>
__global__ void im2col_kernel(const elem_t* x, elem_t* out, uint32_t M,
                              uint32_t Kpad, uint32_t Ktot, uint32_t C,
                              uint32_t H, uint32_t W, uint32_t Q, uint32_t S,
                              int u, int v, int ph, int pw, FastDiv dQ,
                              FastDiv dPQ, FastDiv dC, FastDiv dS) {
  size_t idx = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t total = (size_t)M * Kpad;
  for (; idx < total; idx += (size_t)gridDim.x * blockDim.x) {
    uint32_t m = idx / Kpad, k = idx % Kpad;
    elem_t val = (elem_t)0.f;
    if (k < Ktot) {
      uint32_t n = dPQ.div(m), pq = dPQ.mod(m, n);
      uint32_t p = dQ.div(pq), q = dQ.mod(pq, p);
      uint32_t rs = dC.div(k), c = dC.mod(k, rs);
      uint32_t r = dS.div(rs), s = dS.mod(rs, r);
      int ih = (int)(p * u) - ph + (int)r;
      int iw = (int)(q * v) - pw + (int)s;
      if ((uint32_t)ih < H && (uint32_t)iw < W)
        val = x[(((size_t)n * H + ih) * W + iw) * C + c];
    }
    out[idx] = val;
  }
}

// Weight transpose for dgrad: (K,R,S,C) memory -> (C,R,S,K) memory.
// One thread per 8-k output chunk: writes are dense 16B chunks (a wave emits
// 1 KB contiguous); the 2B-strided reads hit the MALL — conv weights are
// <=5 MB. Replaces at::permute().contiguous(), measured ~250 GB/s there.
template <typename elem_t>
__global__ void transpose_w_crsk_kernel(const elem_t* __restrict__ src,
                                        elem_t* __restrict__ dst, uint32_t C,
                                        uint32_t RS, uint32_t K, uint32_t total,
                                        FastDiv dRSK8, FastDiv dK8) {
  using V8 = typename E8<elem_t>::v8;
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  uint32_t c = dRSK8.div(i);
  uint32_t rem = dRSK8.mod(i, c);
  uint32_t rs = dK8.div(rem);
  uint32_t k0 = dK8.mod(rem, rs) * 8;
  elem_t tmp[8];
#pragma unroll
  for (int j = 0; j < 8; ++j)
    tmp[j] = src[((size_t)(k0 + j) * RS + rs) * C + c];
  *(V8*)(dst + ((size_t)c * RS + rs) * K + k0) = *(V8*)tmp;
}

template <typename elem_t>
__global__ void cast_f32_bf16_kernel(const float* __restrict__ in,
                                     elem_t* __restrict__ out, size_t total8) {
  using V8 = typename E8<elem_t>::v8;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total8; i += stride) {
    V8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) o[e] = (elem_t)in[i * 8 + e];
    *(V8*)(out + i * 8) = o;
  }
}

// drain-and-rezero variant for the persistent wgrad split-K accumulator:
// casting is the only consumer, so zeroing here replaces a separate fill
// before every wgrad launch (the fills were pure launch overhead, ~50/step).
template <typename elem_t>
__global__ void cast_f32_bf16_zero_kernel(float* __restrict__ in,
                                          elem_t* __restrict__ out,
                                          size_t total8) {
  using V8 = typename E8<elem_t>::v8;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total8; i += stride) {
    V8 o;
    f32x4 z4 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int e = 0; e < 8; ++e) o[e] = (elem_t)in[i * 8 + e];
    *(V8*)(out + i * 8) = o;
    *(f32x4*)(in + i * 8) = z4;
    *(f32x4*)(in + i * 8 + 4) = z4;
  }
}

// ============================================================== host side ==

template <typename elem_t>
static const elem_t* zero_page(const at::Tensor& like) {
  // a zeroed 16-bit pattern reads as 0.0 in bf16 and fp16 alike
  static at::Tensor z;
  if (!z.defined() || z.device() != like.device())
    z = at::zeros({64}, like.options().dtype(at::kBFloat16));
  return (const elem_t*)z.data_ptr();
}

static inline uint32_t ceil_div(uint32_t a, uint32_t b) { return (a + b - 1) / b; }

static bool env_flag(const char* name) {
  const char* v = getenv(name);
  return v && v[0] == '1';
}

// Persistent pre-zeroed fp32 accumulator for wgrad split-K atomics, keyed by
// shape. The consumer (cast_f32_bf16_zero_kernel) re-zeroes it in the same
// pass, so the per-call at::zeros fill disappears from the step.
static std::map<std::tuple<int, long, long>, at::Tensor>& wgrad_ws_cache() {
  static std::map<std::tuple<int, long, long>, at::Tensor> cache;
  return cache;
}

static at::Tensor wgrad_acc_ws(uint32_t Ko, uint32_t RSC, const at::Tensor& like) {
  auto& cache = wgrad_ws_cache();
  auto key = std::make_tuple((int)like.get_device(), (long)Ko, (long)RSC);
  auto it = cache.find(key);
  if (it == cache.end())
    it = cache
             .emplace(key, at::zeros({(long)Ko, (long)RSC},
                                     like.options().dtype(at::kFloat)))
             .first;
  return it->second;
}

// debug: max |entry| per cached workspace — all must be 0 between steps
std::vector<std::tuple<long, long, double>> wgrad_ws_stats() {
  std::vector<std::tuple<long, long, double>> out;
  for (auto& kv : wgrad_ws_cache())
    out.emplace_back(std::get<1>(kv.first), std::get<2>(kv.first),
                     kv.second.abs().max().item<double>());
  return out;
}

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

#define CHECK_BF16_CUDA(t)                                                   \
  TORCH_CHECK((t).is_cuda() && ((t).scalar_type() == at::kBFloat16 ||        \
                                (t).scalar_type() == at::kHalf),             \
              #t " must be a CUDA bf16/fp16 tensor")

// route a GEMM through the 256² kernel when the shape qualifies (big M/N,
// pipelined K%64==0, 128-row slab granularity preserved). Dense AND gather
// providers (conv fwd/dgrad A) are supported; OOB k pieces read the zero
// page via the provider's own bounds checks on the gather side.
template <class PA, class PB, class EPI>
static bool try_gemm256(const PA& pa, const PB& pb, const EPI& epi,
                        uint32_t M, uint32_t N, uint32_t K, uint32_t splitk) {
  if constexpr (!EPI::kLdsStage) {
    return false;
  } else {
    static const bool off = env_flag("DTMX_DISABLE_GEMM256");
    if (off || splitk > 1) return false;
    if (M < 512 || N < 192 || K < 128 || K % 64 != 0) return false;
    uint32_t tiles_m = ceil_div(M, 256), tiles_n = ceil_div(N, 256);
    if (tiles_m * tiles_n < 160) return false;  // underfilled grid
    dim3 grid(tiles_m * tiles_n);
    gemm256f_kernel<PA, PB, EPI><<<grid, 512, 0, cur_stream()>>>(
        pa, pb, epi, K / 64, tiles_n);
    return true;
  }
}

// small grids underfill 256 CUs: callers pass want_splitk=true to let the
// launcher split K into an fp32 atomic buffer (the caller casts back).
template <class PA, class PB, class EPI>
static void launch_gemm(const PA& pa, const PB& pb, const EPI& epi, uint32_t M,
                        uint32_t N, uint32_t K, uint32_t splitk = 1) {
  if (try_gemm256(pa, pb, epi, M, N, K, splitk)) return;
  uint32_t ktiles_total = ceil_div(K, 64);
  splitk = std::min(splitk, ktiles_total);
  uint32_t kt_per = ceil_div(ktiles_total, splitk);
  uint32_t tiles_m = ceil_div(M, 128);
  if (N <= 64) {  // narrow-N tile: BN=64
    uint32_t tiles_n = ceil_div(N, 64);
    dim3 grid(tiles_m * tiles_n, ceil_div(ktiles_total, kt_per));
    gemm_tn_kernel<2, PA, PB, EPI><<<grid, 256, 0, cur_stream()>>>(
        pa, pb, epi, ktiles_total, tiles_n, kt_per);
  } else {
    uint32_t tiles_n = ceil_div(N, 128);
    dim3 grid(tiles_m * tiles_n, ceil_div(ktiles_total, kt_per));
    gemm_tn_kernel<4, PA, PB, EPI><<<grid, 256, 0, cur_stream()>>>(
        pa, pb, epi, ktiles_total, tiles_n, kt_per);
  }
}

template <class PA, class PB, class EPI>
static void launch_gemm_nt(const PA& pa, const PB& pb, const EPI& epi,
                           uint32_t M, uint32_t N, uint32_t K,
                           uint32_t splitk = 1) {
  uint32_t ktiles_total = ceil_div(K, 64);
  splitk = std::min(splitk, ktiles_total);
  uint32_t kt_per = ceil_div(ktiles_total, splitk);
  // measured: depth 3 drops occupancy to 1 block/CU (96 KiB LDS at NJ=4)
  // and LOSES 20-35% on every wgrad shape — the double buffer's 2-block
  // overlap already covers the HBM latency better than a deeper ring at
  // half the resident waves. Default 2; DTMX_NT_DEPTH=3 for A/B.
  static const bool depth3 = [] {
    const char* v = getenv("DTMX_NT_DEPTH");
    return v && v[0] == '3';
  }();
  // XCD-grouped slice scheduling (see gemm_nt_kernel GROUP doc): 1-D grid,
  // all tiles of a split-K slice on one die so shared dy K-tiles hit L2.
  // Measured per-shape (tools/bench_wgrad.py, bs1024): +11-12% at T=9 tiles
  // (128-channel 3x3 wgrad class, slices/XCD fill 63 of 64 slots), -3% at
  // T=144, -33%/-43% at T=36/T=3 (slot spill -> straggler slices, and an
  // unexplained loss on the flat WM1 tile) — so auto mode gates it to the
  // tile-count class that wins. DTMX_NT_GROUP=0/1 forces off/on everywhere.
  static const int group_env = [] {
    const char* v = getenv("DTMX_NT_GROUP");
    return v ? (v[0] == '1' ? 1 : 0) : -1;
  }();
  uint32_t tiles_mn_probe =
      (M <= 64 && N > 64) ? ceil_div(N, 256)
                          : ceil_div(M, 128) * ceil_div(N, N <= 64 ? 64u : 128u);
  const bool group = group_env == 1 ||
                     (group_env == -1 && M > 64 && tiles_mn_probe > 4 &&
                      tiles_mn_probe <= 16);
  uint32_t nslices = ceil_div(ktiles_total, kt_per);
  if (group) {
    // slice->XCD is slice%8: a slice count that is not a multiple of 8
    // leaves some dies with one more slice than others (measured -40% on
    // the 3-slice 512-channel wgrad) — round UP to 8k slices instead
    uint32_t ns8 = std::min(ktiles_total, ceil_div(nslices, 8u) * 8u);
    kt_per = ceil_div(ktiles_total, ns8);
    nslices = ceil_div(ktiles_total, kt_per);
  }
  if (M <= 64 && N > 64) {
    // flat 64x256 tile: ResNet's K_out=64 wgrads waste half the 128-row
    // tile on zero rows (see gemm_nt_kernel WM doc)
    // DTMX_NT_WM1NJ2: 64x128 flat tile (48 KiB LDS -> 3 blocks/CU, 12
    // waves) — occupancy-vs-re-read probe for this latency-bound class
    static const bool wm1nj2 = [] {
      const char* v = getenv("DTMX_NT_WM1NJ2");
      return v && v[0] == '1';
    }();
    if (wm1nj2) {
      uint32_t tiles_n = ceil_div(N, 128);
      dim3 grid(tiles_n, nslices);
      gemm_nt_kernel<2, 1, 2, PA, PB, EPI><<<grid, 256, 0, cur_stream()>>>(
          pa, pb, epi, ktiles_total, tiles_n, kt_per);
      return;
    }
    uint32_t tiles_n = ceil_div(N, 256);
    dim3 grid(tiles_n, nslices);
    if (group && !depth3)
      gemm_nt_kernel<4, 1, 2, PA, PB, EPI, 1>
          <<<8 * tiles_n * ceil_div(nslices, 8u), 256, 0, cur_stream()>>>(
              pa, pb, epi, ktiles_total, tiles_n, kt_per, tiles_n);
    else if (depth3)
      gemm_nt_kernel<4, 1, 3, PA, PB, EPI><<<grid, 256, 0, cur_stream()>>>(
          pa, pb, epi, ktiles_total, tiles_n, kt_per);
    else
      gemm_nt_kernel<4, 1, 2, PA, PB, EPI><<<grid, 256, 0, cur_stream()>>>(
          pa, pb, epi, ktiles_total, tiles_n, kt_per);
    return;
  }
  uint32_t tiles_m = ceil_div(M, 128);
  if (N <= 64) {
    uint32_t tiles_n = ceil_div(N, 64);
    dim3 grid(tiles_m * tiles_n, nslices);
    if (group && !depth3)
      gemm_nt_kernel<2, 2, 2, PA, PB, EPI, 1>
          <<<8 * tiles_m * tiles_n * ceil_div(nslices, 8u), 256, 0,
             cur_stream()>>>(pa, pb, epi, ktiles_total, tiles_n, kt_per,
                             tiles_m * tiles_n);
    else if (depth3)
      gemm_nt_kernel<2, 2, 3, PA, PB, EPI><<<grid, 256, 0, cur_stream()>>>(
          pa, pb, epi, ktiles_total, tiles_n, kt_per);
    else
      gemm_nt_kernel<2, 2, 2, PA, PB, EPI><<<grid, 256, 0, cur_stream()>>>(
          pa, pb, epi, ktiles_total, tiles_n, kt_per);
  } else {
    uint32_t tiles_n = ceil_div(N, 128);
    dim3 grid(tiles_m * tiles_n, nslices);
    if (group && !depth3)
      gemm_nt_kernel<4, 2, 2, PA, PB, EPI, 1>
          <<<8 * tiles_m * tiles_n * ceil_div(nslices, 8u), 256, 0,
             cur_stream()>>>(pa, pb, epi, ktiles_total, tiles_n, kt_per,
                             tiles_m * tiles_n);
    else if (depth3)
      gemm_nt_kernel<4, 2, 3, PA, PB, EPI><<<grid, 256, 0, cur_stream()>>>(
          pa, pb, epi, ktiles_total, tiles_n, kt_per);
    else
      gemm_nt_kernel<4, 2, 2, PA, PB, EPI><<<grid, 256, 0, cur_stream()>>>(
          pa, pb, epi, ktiles_total, tiles_n, kt_per);
  }
}

// -- plain GEMM entry points (FullyConnected; reference fully_connected-inl.h)

static at::Tensor pad_cols8(const at::Tensor& t) {
  // zero-pad the last dim to a multiple of 8 (k-piece granularity)
  long k = t.size(-1);
  if (k % 8 == 0) return t.contiguous();
  return at::constant_pad_nd(t, {0, 8 - (k % 8)}, 0.0).contiguous();
}

at::Tensor linear_fwd(const at::Tensor& x, const at::Tensor& w,
                      const c10::optional<at::Tensor>& bias) {
  DTMX_DISPATCH_16(x.scalar_type(), "linear_fwd", {
    CHECK_BF16_CUDA(x);
    CHECK_BF16_CUDA(w);
    TORCH_CHECK(x.size(1) == w.size(1), "linear: in_features mismatch (x ",
                x.size(1), " vs w ", w.size(1), ")");
    auto xc = pad_cols8(x);
    auto wc = pad_cols8(w);
    uint32_t M = xc.size(0), K = xc.size(1), N = wc.size(0);
    auto y = at::empty({(long)M, (long)N}, x.options());
    at::Tensor bias_f;
    const float* bp = nullptr;
    if (bias.has_value()) {
      bias_f = bias->to(at::kFloat).contiguous();
      bp = bias_f.data_ptr<float>();
    }
    DenseP<elem_t> pa{(const elem_t*)xc.data_ptr(), zero_page<elem_t>(x), M, K, K};
    DenseP<elem_t> pb{(const elem_t*)wc.data_ptr(), zero_page<elem_t>(x), N, K, K};
    EpiBF16<elem_t> epi{(elem_t*)y.data_ptr(), bp, M, N, 0};
    launch_gemm(pa, pb, epi, M, N, K);
    return y;

  });
  return at::Tensor();
}

template <typename elem_t>
static at::Tensor transpose2d(const at::Tensor& in, uint32_t out_ld_pad = 0) {
  // [M][C] -> [C][Mpad64?]: out_ld defaults to M rounded to 8
  CHECK_BF16_CUDA(in);
  auto inc = in.contiguous();
  uint32_t M = inc.size(0), C = inc.size(1);
  uint32_t out_ld = out_ld_pad ? out_ld_pad : ((M + 7) / 8) * 8;
  TORCH_CHECK(C % 8 == 0, "transpose2d: cols must be a multiple of 8");
  auto out = at::empty({(long)C, (long)out_ld}, in.options());
  uint32_t tiles_m = ceil_div(out_ld, 64), tiles_c = ceil_div(C, 64);
  IdentityRows<elem_t> rows{(const elem_t*)inc.data_ptr(), M, C};
  transpose_rowgather_kernel<IdentityRows<elem_t>>
      <<<tiles_m * tiles_c, 256, 0, cur_stream()>>>(
          rows, (elem_t*)out.data_ptr(), C, out_ld, 0, tiles_m);
  return out;
}

at::Tensor linear_dgrad(const at::Tensor& dy, const at::Tensor& w) {
  DTMX_DISPATCH_16(dy.scalar_type(), "linear_dgrad", {
    CHECK_BF16_CUDA(dy);
    uint32_t M = dy.size(0), N = dy.size(1), K = w.size(1);
    uint32_t Npad = ((N + 7) / 8) * 8;
    auto dyc = pad_cols8(dy);                      // [M][Npad]
    auto wt = transpose2d<elem_t>(pad_cols8(w), Npad);     // [Kpad8][Npad] (rows>N zero)
    auto dx = at::empty({(long)M, (long)K}, dy.options());
    DenseP<elem_t> pa{(const elem_t*)dyc.data_ptr(), zero_page<elem_t>(dy), M, Npad, Npad};
    DenseP<elem_t> pb{(const elem_t*)wt.data_ptr(), zero_page<elem_t>(dy), K, Npad, Npad};
    EpiBF16<elem_t> epi{(elem_t*)dx.data_ptr(), nullptr, M, K, 0};
    launch_gemm(pa, pb, epi, M, K, Npad);
    return dx;

  });
  return at::Tensor();
}

at::Tensor linear_wgrad(const at::Tensor& dy, const at::Tensor& x) {
  DTMX_DISPATCH_16(dy.scalar_type(), "linear_wgrad", {
    CHECK_BF16_CUDA(dy);
    uint32_t M = dy.size(0), N = dy.size(1), K = x.size(1);
    uint32_t Mpad = ((M + 7) / 8) * 8;
    auto dyt = transpose2d<elem_t>(pad_cols8(dy), Mpad);  // [Npad][Mpad]
    auto xt = transpose2d<elem_t>(pad_cols8(x), Mpad);    // [Kpad][Mpad]
    auto dw = at::empty({(long)N, (long)K}, dy.options());
    DenseP<elem_t> pa{(const elem_t*)dyt.data_ptr(), zero_page<elem_t>(dy), N, Mpad, Mpad};
    DenseP<elem_t> pb{(const elem_t*)xt.data_ptr(), zero_page<elem_t>(dy), K, Mpad, Mpad};
    EpiBF16<elem_t> epi{(elem_t*)dw.data_ptr(), nullptr, N, K, 0};
    launch_gemm(pa, pb, epi, N, K, Mpad);
    return dw;

  });
  return at::Tensor();
}

// ------------------------------------------------------------- conv fwd

static void conv_out_dims(uint32_t H, uint32_t W, uint32_t R, uint32_t S,
                          int stride, int pad, uint32_t& P, uint32_t& Q) {
  P = (H + 2 * pad - R) / stride + 1;
  Q = (W + 2 * pad - S) / stride + 1;
}

// run a bf16-output GEMM through fp32 atomics + cast when its natural grid
// underfills the chip (e.g. the 7x7 resnet stage: 196 blocks on 256 CUs).
template <class PA, class PB>
static bool smallgrid_splitk(const PA& pa, const PB& pb, at::Tensor& out_bf16,
                             uint32_t M, uint32_t N, uint32_t K) {
  using elem_t = typename PA::elem;
  uint32_t tiles = ceil_div(M, 128) * ceil_div(N, N <= 64 ? 64 : 128);
  uint32_t ktiles = ceil_div(K, 64);
  // only for severely underfilled grids: the fp32-atomic + cast overhead and
  // the shortened per-block K loop beat plain launch only below ~160 blocks
  // (measured: 392-block l3 conv lost 40%, 196-block l4 was a wash).
  if (tiles >= 160 || ktiles < 8) return false;
  uint32_t splitk = std::min(ktiles / 4, std::max<uint32_t>(2, 1024 / tiles));
  auto acc = at::zeros({(long)M, (long)N}, out_bf16.options().dtype(at::kFloat));
  EpiAtomicF32<elem_t> epi{acc.data_ptr<float>(), M, N};
  launch_gemm(pa, pb, epi, M, N, K, splitk);
  size_t t8 = (size_t)M * N / 8;  // N % 8 == 0 everywhere this is used
  cast_f32_bf16_kernel<<<std::min<size_t>((t8 + 255) / 256, 2048), 256, 0,
                         cur_stream()>>>(acc.data_ptr<float>(),
                                         (elem_t*)out_bf16.data_ptr(), t8);
  return true;
}

at::Tensor conv_fwd(const at::Tensor& x, const at::Tensor& w, long stride,
                    long pad) {
  DTMX_DISPATCH_16(x.scalar_type(), "conv_fwd", {
    CHECK_BF16_CUDA(x);
    CHECK_BF16_CUDA(w);
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "x must be NHWC");
    TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast), "w must be KRSC");
    uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W_ = x.size(3);
    uint32_t Ko = w.size(0), R = w.size(2), S = w.size(3);
    uint32_t P, Q;
    conv_out_dims(H, W_, R, S, stride, pad, P, Q);
    auto y = at::empty({(long)N, (long)Ko, (long)P, (long)Q},
                       x.options(), at::MemoryFormat::ChannelsLast);
    uint32_t M = N * P * Q, Ktot = R * S * C;
    EpiBF16<elem_t> epi{(elem_t*)y.data_ptr(), nullptr, M, Ko, 0};
    if (R == 1 && S == 1 && stride == 1 && pad == 0 && C % 8 == 0) {
      // 1x1/s1: the im2col matrix IS x — pure dense GEMM, no gather decode
      DenseP<elem_t> pa{(const elem_t*)x.data_ptr(), zero_page<elem_t>(x), M, C, C};
      DenseP<elem_t> pb{(const elem_t*)w.data_ptr(), zero_page<elem_t>(x), Ko, C, C};
      if (!smallgrid_splitk(pa, pb, y, M, Ko, C))
        launch_gemm(pa, pb, epi, M, Ko, C);
    } else if (C % 8 == 0) {
      DenseP<elem_t> pb{(const elem_t*)w.data_ptr(), zero_page<elem_t>(x), Ko, Ktot, Ktot};
      ConvFwdA<elem_t> pa;
      pa.x = (const elem_t*)x.data_ptr();
      pa.zero = zero_page<elem_t>(x);
      pa.M = M; pa.Ktot = Ktot; pa.C = C; pa.H = H; pa.W = W_; pa.Q = Q; pa.S = S;
      pa.u = stride; pa.v = stride; pa.ph = pad; pa.pw = pad;
      pa.dQ.init(Q); pa.dPQ.init(P * Q); pa.dC.init(C); pa.dS.init(S);
      if (!smallgrid_splitk(pa, pb, y, M, Ko, Ktot))
        launch_gemm(pa, pb, epi, M, Ko, Ktot);
    } else {
      // small-C path (3-channel stem): materialized im2col, then dense GEMM.
      uint32_t Kpad = ((Ktot + 63) / 64) * 64;
      auto col = at::empty({(long)M, (long)Kpad}, x.options());
      FastDiv dQ, dPQ, dC, dS;
      dQ.init(Q); dPQ.init(P * Q); dC.init(C); dS.init(S);
      size_t total = (size_t)M * Kpad;
      uint32_t blocks = std::min<size_t>((total + 255) / 256, 16384);
      im2col_kernel<<<blocks, 256, 0, cur_stream()>>>(
          (const elem_t*)x.data_ptr(), (elem_t*)col.data_ptr(), M, Kpad, Ktot, C,
          H, W_, Q, S, stride, stride, pad, pad, dQ, dPQ, dC, dS);
      // zero-pad the weight rows to Kpad so the k tail multiplies 0*0, not 0*NaN
      auto wpad = at::constant_pad_nd(
          w.permute({0, 2, 3, 1}).reshape({(long)Ko, (long)Ktot}),
          {0, (long)(Kpad - Ktot)}, 0.0).contiguous();
      DenseP<elem_t> pa{(const elem_t*)col.data_ptr(), zero_page<elem_t>(x), M, Kpad, Kpad};
      DenseP<elem_t> pb{(const elem_t*)wpad.data_ptr(), zero_page<elem_t>(x), Ko, Kpad, Kpad};
      launch_gemm(pa, pb, epi, M, Ko, Kpad);
    }
    return y;

  });
  return at::Tensor();
}

// conv forward with fused BN statistics: returns (y, psum, psumsq) where the
// slabs are [tiles_m][Ko] per-block column sums/sumsq of y (epilogue fusion —
// the following BatchNorm skips its whole-tensor stats read). Returns empty
// slabs when the shape routed through a non-fusable path; the caller falls
// back to the standalone stats kernel.
std::vector<at::Tensor> conv_fwd_stats(const at::Tensor& x, const at::Tensor& w,
                                       long stride, long pad) {
  DTMX_DISPATCH_16(x.scalar_type(), "conv_fwd_stats", {
    CHECK_BF16_CUDA(x);
    TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "x must be NHWC");
    uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W_ = x.size(3);
    uint32_t Ko = w.size(0), R = w.size(2), S = w.size(3);
    uint32_t P, Q;
    conv_out_dims(H, W_, R, S, stride, pad, P, Q);
    uint32_t M = N * P * Q, Ktot = R * S * C;
    uint32_t tiles_m = ceil_div(M, 128);
    uint32_t tiles = tiles_m * ceil_div(Ko, Ko <= 64 ? 64 : 128);
    uint32_t ktiles = ceil_div(Ktot, 64);
    bool smallgrid = tiles < 160 && ktiles >= 8;  // mirrors smallgrid_splitk
    if (C % 8 != 0 || smallgrid) {
      auto y = conv_fwd(x, w, stride, pad);
      auto empty = at::empty({0}, x.options().dtype(at::kFloat));
      return {y, empty, empty};
    }
    auto y = at::empty({(long)N, (long)Ko, (long)P, (long)Q}, x.options(),
                       at::MemoryFormat::ChannelsLast);
    auto opt_f = x.options().dtype(at::kFloat);
    // every slab element is written exactly once by its owning block
    auto psum = at::empty({(long)tiles_m, (long)Ko}, opt_f);
    auto psumsq = at::empty({(long)tiles_m, (long)Ko}, opt_f);
    EpiBF16FwdStats<elem_t> epi;
    epi.c = (elem_t*)y.data_ptr();
    epi.bn_psum = psum.data_ptr<float>();
    epi.bn_psumsq = psumsq.data_ptr<float>();
    epi.M = M;
    epi.N = Ko;
    if (R == 1 && S == 1 && stride == 1 && pad == 0) {
      DenseP<elem_t> pa{(const elem_t*)x.data_ptr(), zero_page<elem_t>(x), M, C, C};
      DenseP<elem_t> pb{(const elem_t*)w.data_ptr(), zero_page<elem_t>(x), Ko, C, C};
      launch_gemm(pa, pb, epi, M, Ko, C);
    } else {
      DenseP<elem_t> pb{(const elem_t*)w.data_ptr(), zero_page<elem_t>(x), Ko, Ktot, Ktot};
      ConvFwdA<elem_t> pa;
      pa.x = (const elem_t*)x.data_ptr();
      pa.zero = zero_page<elem_t>(x);
      pa.M = M; pa.Ktot = Ktot; pa.C = C; pa.H = H; pa.W = W_; pa.Q = Q; pa.S = S;
      pa.u = stride; pa.v = stride; pa.ph = pad; pa.pw = pad;
      pa.dQ.init(Q); pa.dPQ.init(P * Q); pa.dC.init(C); pa.dS.init(S);
      launch_gemm(pa, pb, epi, M, Ko, Ktot);
    }
    return {y, psum, psumsq};
  });
  return {};
}

// ------------------------------------------------------------- conv dgrad

at::Tensor conv_dgrad(const at::Tensor& dy, const at::Tensor& w, long stride,
                      long pad, long H, long W_,
                      const c10::optional<at::Tensor>& acc) {
  DTMX_DISPATCH_16(dy.scalar_type(), "conv_dgrad", {
    CHECK_BF16_CUDA(dy);
    TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast), "dy must be NHWC");
    uint32_t N = dy.size(0), Ko = dy.size(1), P = dy.size(2), Q = dy.size(3);
    uint32_t C = w.size(1), R = w.size(2), S = w.size(3);
    // pad the out-channel (k-piece) dim to a multiple of 8 when needed
    at::Tensor dyk = dy.permute({0, 2, 3, 1});        // NHWC view (contiguous)
    at::Tensor wtk = w.permute({1, 2, 3, 0});          // (C,R,S,K)
    if (Ko % 8) {
      long padk = 8 - (Ko % 8);
      dyk = at::constant_pad_nd(dyk, {0, padk}, 0.0);
      wtk = at::constant_pad_nd(wtk, {0, padk}, 0.0);
      Ko += padk;
    }
    auto dyc = dyk.contiguous();
    // W^T in (C,R,S,Ko) dense layout — custom kernel when w is the plain
    // channels_last parameter (the training path); at::copy otherwise
    at::Tensor wt;
    static const bool no_wtk = env_flag("DTMX_DISABLE_WT_KERNEL");
    if (!no_wtk && Ko == (uint32_t)w.size(0) && Ko % 8 == 0 &&
        w.is_contiguous(at::MemoryFormat::ChannelsLast)) {
      uint32_t RS = R * S, K8 = Ko / 8, total = C * RS * K8;
      wt = at::empty({(long)C, (long)R, (long)S, (long)Ko}, dy.options());
      FastDiv dRSK8, dK8;
      dRSK8.init(RS * K8);
      dK8.init(K8);
      transpose_w_crsk_kernel<<<ceil_div(total, 256), 256, 0, cur_stream()>>>(
          (const elem_t*)w.data_ptr(), (elem_t*)wt.data_ptr(), C, RS, Ko, total,
          dRSK8, dK8);
    } else {
      wt = wtk.contiguous();
    }
    // residual-join fusion: dx = gemm + acc, accumulated IN PLACE into acc
    // (one read replaces the separate read+read+write grad-sum at the fork)
    const bool have_acc = acc.has_value() && acc->defined();
    at::Tensor dx;
    if (have_acc) {
      TORCH_CHECK(acc->is_contiguous(at::MemoryFormat::ChannelsLast) &&
                      acc->sizes() == at::IntArrayRef({(long)N, (long)C,
                                                       (long)H, (long)W_}),
                  "conv_dgrad: acc must be NHWC of the dx shape");
      dx = *acc;
    } else {
      dx = at::empty({(long)N, (long)C, (long)H, (long)W_}, dy.options(),
                     at::MemoryFormat::ChannelsLast);
    }
    uint32_t M = N * H * W_, Ktot = R * S * Ko;
    if (R == 1 && S == 1 && pad == 0 && C % 8 == 0) {
      // 1x1 dgrad: dense dy @ W^T. Stride 1 writes rows directly; stride u>1
      // scatters row (n,p,q) to pixel (n, p*u, q*v) of a zeroed dx — the
      // gathered formulation wastes 1-1/u^2 of its blocks on all-zero rows
      // (measured 107 TF vs ~300 dense).
      uint32_t Mn = N * P * Q;
      DenseP<elem_t> pad_{(const elem_t*)dyc.data_ptr(), zero_page<elem_t>(dy), Mn, Ko, Ko};
      DenseP<elem_t> pbd{(const elem_t*)wt.data_ptr(), zero_page<elem_t>(dy), C, Ko, Ko};
      if (stride == 1) {
        EpiBF16<elem_t> epid{(elem_t*)dx.data_ptr(), nullptr, Mn, C, 0};
        if (have_acc) epid.acc = (const elem_t*)dx.data_ptr();
        if (have_acc || !smallgrid_splitk(pad_, pbd, dx, Mn, C, Ko))
          launch_gemm(pad_, pbd, epid, Mn, C, Ko);
      } else {
        if (!have_acc) dx.zero_();
        EpiBF16Scatter<elem_t> epis;
        epis.dx = (elem_t*)dx.data_ptr();
        epis.M = Mn; epis.N = C; epis.H = H; epis.W = W_; epis.Q = Q;
        epis.u = stride; epis.v = stride;
        epis.accumulate = have_acc ? 1 : 0;
        epis.dQ.init(Q); epis.dPQ.init(P * Q);
        launch_gemm(pad_, pbd, epis, Mn, C, Ko);
      }
      return dx;
    }
    ConvDgradA<elem_t> pa;
    pa.dy = (const elem_t*)dyc.data_ptr();
    pa.zero = zero_page<elem_t>(dy);
    pa.M = M; pa.Ktot = Ktot; pa.Ko = Ko; pa.H = H; pa.W = W_; pa.P = P; pa.Q = Q;
    pa.S = S; pa.u = stride; pa.v = stride; pa.ph = pad; pa.pw = pad;
    pa.dW_.init(W_); pa.dHW.init(H * W_); pa.dKo.init(Ko); pa.dS.init(S);
    DenseP<elem_t> pb{(const elem_t*)wt.data_ptr(), zero_page<elem_t>(dy), C, Ktot, Ktot};
    EpiBF16<elem_t> epi{(elem_t*)dx.data_ptr(), nullptr, M, C, 0};
    if (have_acc) epi.acc = (const elem_t*)dx.data_ptr();
    if (have_acc || !smallgrid_splitk(pa, pb, dx, M, C, Ktot))
      launch_gemm(pa, pb, epi, M, C, Ktot);
    return dx;

  });
  return at::Tensor();
}

// conv dgrad with the BN-backward epilogue fusion (EpiBnBwd): computes
// dy_bn = conv_dgrad(dy, w) [+ acc], relu-masks it with y, writes the masked
// gradient and [tiles_m][C] partial (Σg, Σg*xhat) slabs. Returns
// {g, pdb, pdg}; finish with bn_bwd_finalize_slabs + bn_bwd_dx_presummed.
// Not applicable to the 1x1/stride>1 scatter path (the scatter epilogue
// covers only 1/u^2 of the output pixels).
std::vector<at::Tensor> conv_dgrad_bnfuse(
    const at::Tensor& dy, const at::Tensor& w, long stride, long pad, long H,
    long W_, const c10::optional<at::Tensor>& acc, const at::Tensor& y,
    const at::Tensor& xin, const at::Tensor& mean, const at::Tensor& invstd) {
  DTMX_DISPATCH_16(dy.scalar_type(), "conv_dgrad_bnfuse", {
    CHECK_BF16_CUDA(dy);
    TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast), "dy must be NHWC");
    TORCH_CHECK(y.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                    xin.is_contiguous(at::MemoryFormat::ChannelsLast),
                "bnfuse: y/xin must be NHWC");
    uint32_t N = dy.size(0), Ko = dy.size(1), P = dy.size(2), Q = dy.size(3);
    uint32_t C = w.size(1), R = w.size(2), S = w.size(3);
    TORCH_CHECK(!(R == 1 && S == 1 && stride > 1),
                "conv_dgrad_bnfuse: strided 1x1 (scatter) is not fusable");
    TORCH_CHECK(C % 8 == 0, "conv_dgrad_bnfuse: C must be a multiple of 8");
    at::Tensor dyk = dy.permute({0, 2, 3, 1});
    at::Tensor wtk = w.permute({1, 2, 3, 0});
    if (Ko % 8) {
      long padk = 8 - (Ko % 8);
      dyk = at::constant_pad_nd(dyk, {0, padk}, 0.0);
      wtk = at::constant_pad_nd(wtk, {0, padk}, 0.0);
      Ko += padk;
    }
    auto dyc = dyk.contiguous();
    at::Tensor wt;
    if (Ko == (uint32_t)w.size(0) && Ko % 8 == 0 &&
        w.is_contiguous(at::MemoryFormat::ChannelsLast)) {
      uint32_t RS = R * S, K8 = Ko / 8, total = C * RS * K8;
      wt = at::empty({(long)C, (long)R, (long)S, (long)Ko}, dy.options());
      FastDiv dRSK8, dK8;
      dRSK8.init(RS * K8);
      dK8.init(K8);
      transpose_w_crsk_kernel<<<ceil_div(total, 256), 256, 0, cur_stream()>>>(
          (const elem_t*)w.data_ptr(), (elem_t*)wt.data_ptr(), C, RS, Ko, total,
          dRSK8, dK8);
    } else {
      wt = wtk.contiguous();
    }
    const bool have_acc = acc.has_value() && acc->defined();
    at::Tensor dx;
    if (have_acc) {
      dx = *acc;
    } else {
      dx = at::empty({(long)N, (long)C, (long)H, (long)W_}, dy.options(),
                     at::MemoryFormat::ChannelsLast);
    }
    uint32_t M = N * H * W_, Ktot = R * S * Ko;
    uint32_t tiles_m = ceil_div(M, 128);
    auto opt_f = dy.options().dtype(at::kFloat);
    // every slab element is written exactly once by its owning block
    auto pdb = at::empty({(long)tiles_m, (long)C}, opt_f);
    auto pdg = at::empty({(long)tiles_m, (long)C}, opt_f);
    EpiBnBwd<elem_t> epi;
    epi.c = (elem_t*)dx.data_ptr();
    epi.acc = have_acc ? (const elem_t*)dx.data_ptr() : nullptr;
    epi.bnb_y = (const elem_t*)y.data_ptr();
    epi.bnb_x = (const elem_t*)xin.data_ptr();
    epi.bnb_mean = mean.data_ptr<float>();
    epi.bnb_invstd = invstd.data_ptr<float>();
    epi.bnb_pdb = pdb.data_ptr<float>();
    epi.bnb_pdg = pdg.data_ptr<float>();
    epi.M = M;
    epi.N = C;
    if (R == 1 && S == 1 && pad == 0) {  // stride==1 guaranteed above
      DenseP<elem_t> pa{(const elem_t*)dyc.data_ptr(), zero_page<elem_t>(dy),
                        M, Ko, Ko};
      DenseP<elem_t> pb{(const elem_t*)wt.data_ptr(), zero_page<elem_t>(dy),
                        C, Ko, Ko};
      launch_gemm(pa, pb, epi, M, C, Ko);
    } else {
      ConvDgradA<elem_t> pa;
      pa.dy = (const elem_t*)dyc.data_ptr();
      pa.zero = zero_page<elem_t>(dy);
      pa.M = M; pa.Ktot = Ktot; pa.Ko = Ko; pa.H = H; pa.W = W_; pa.P = P;
      pa.Q = Q; pa.S = S; pa.u = stride; pa.v = stride; pa.ph = pad; pa.pw = pad;
      pa.dW_.init(W_); pa.dHW.init(H * W_); pa.dKo.init(Ko); pa.dS.init(S);
      DenseP<elem_t> pb{(const elem_t*)wt.data_ptr(), zero_page<elem_t>(dy),
                        C, Ktot, Ktot};
      launch_gemm(pa, pb, epi, M, C, Ktot);
    }
    return {dx, pdb, pdg};
  });
  return {};
}

// ------------------------------------------------------------- conv wgrad
// dW[ko][(r,s,c)] = sum_{n,p,q} dy[n,p,q,ko] * x[n,pu-ph+r,qv-pw+s,c]
// as TN GEMM over Kd = NPQ with transposed operands + split-K fp32 atomics.

at::Tensor conv_wgrad(const at::Tensor& x, const at::Tensor& dy, long R, long S,
                      long stride, long pad) {
  DTMX_DISPATCH_16(x.scalar_type(), "conv_wgrad", {
    CHECK_BF16_CUDA(x);
    CHECK_BF16_CUDA(dy);
    uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W_ = x.size(3);
    uint32_t Ko = dy.size(1), P = dy.size(2), Q = dy.size(3);
    uint32_t M = N * P * Q;               // the contraction length
    uint32_t Mpad = ((M + 7) / 8) * 8;
    uint32_t RSC = R * S * C;

    if (C % 8 == 0 && Ko % 8 == 0) {
      // transpose-free: dy and the im2col view of x are read NHWC-native by
      // the contraction-major (NT) kernel
      uint32_t tiles_mn =
          (Ko <= 64 && RSC > 64)
              ? ceil_div(RSC, 256)  // flat 64x256 tile (launch_gemm_nt)
              : ceil_div(Ko, 128) * ceil_div(RSC, RSC <= 64 ? 64 : 128);
      uint32_t ktiles = ceil_div(M, 64);
      static const uint32_t target_blocks = [] {
        const char* v = getenv("DTMX_WGRAD_TARGET_BLOCKS");
        // 512 = exactly 2 blocks/CU: the sweep 256..4096 measured 512 best
        // on every wgrad shape (+25% over the old 1024 — no partial block
        // wave, fewest split-K atomic passes that still fill the chip)
        return v ? (uint32_t)atoi(v) : 512u;
      }();
      uint32_t splitk = std::max<uint32_t>(
          1, std::min<uint32_t>(ktiles, target_blocks / std::max(1u, tiles_mn)));
      static const bool no_ws = env_flag("DTMX_DISABLE_WGRAD_WS");
      auto dw32 = no_ws ? at::zeros({(long)Ko, (long)RSC},
                                    x.options().dtype(at::kFloat))
                        : wgrad_acc_ws(Ko, RSC, x);  // pre-zeroed persistent
      WgradDyA<elem_t> pa;
      pa.dy = (const elem_t*)dy.data_ptr();
      pa.zero = zero_page<elem_t>(x);
      pa.Kd = M;
      pa.Mdim = Ko;
      WgradXcolB<elem_t> pb;
      pb.x = (const elem_t*)x.data_ptr();
      pb.zero = zero_page<elem_t>(x);
      pb.Kd = M; pb.Ndim = RSC; pb.C = C; pb.H = H; pb.W = W_; pb.Q = Q; pb.S = S;
      pb.u = stride; pb.v = stride; pb.ph = pad; pb.pw = pad;
      pb.dQ.init(Q); pb.dPQ.init(P * Q); pb.dC.init(C); pb.dS.init(S);
      EpiAtomicF32<elem_t> epi{dw32.data_ptr<float>(), Ko, RSC};
      launch_gemm_nt(pa, pb, epi, Ko, RSC, M, splitk);
      if (no_ws) {
        auto dwo = dw32.reshape({(long)Ko, (long)R, (long)S, (long)C})
                       .to(x.scalar_type())
                       .permute({0, 3, 1, 2});
        return dwo.contiguous(at::MemoryFormat::ChannelsLast);
      }
      // drain to 16-bit and re-zero the workspace for its next use
      auto dwm = at::empty({(long)Ko, (long)R, (long)S, (long)C}, x.options());
      size_t t8 = (size_t)Ko * RSC / 8;
      cast_f32_bf16_zero_kernel<<<std::min<size_t>((t8 + 255) / 256, 2048), 256,
                                  0, cur_stream()>>>(
          dw32.data_ptr<float>(), (elem_t*)dwm.data_ptr(), t8);
      return dwm.permute({0, 3, 1, 2});  // logical (K,C,R,S), channels_last
    }

    // dy^T : [NPQ][Ko] -> [Ko(+pad)][Mpad]
    auto dyt = transpose2d<elem_t>(
        pad_cols8(dy.permute({0, 2, 3, 1}).reshape({(long)M, (long)Ko})), Mpad);

    // gathered im2col^T: [RSC][Mpad] (one transpose-gather launch per (r,s))
    at::Tensor xt;
    if (C % 8 == 0) {
      xt = at::empty({(long)RSC, (long)Mpad}, x.options());
      uint32_t tiles_m = ceil_div(Mpad, 64), tiles_c = ceil_div(C, 64);
      for (uint32_t r = 0; r < (uint32_t)R; ++r)
        for (uint32_t s = 0; s < (uint32_t)S; ++s) {
          Im2colRows<elem_t> rows;
          rows.x = (const elem_t*)x.data_ptr();
          rows.M = M; rows.C = C; rows.H = H; rows.W = W_; rows.Q = Q;
          rows.u = stride; rows.v = stride; rows.ph = pad; rows.pw = pad;
          rows.r = r; rows.s = s;
          rows.dQ.init(Q); rows.dPQ.init(P * Q);
          transpose_rowgather_kernel<Im2colRows<elem_t>>
              <<<tiles_m * tiles_c, 256, 0, cur_stream()>>>(
                  rows, (elem_t*)xt.data_ptr(), C, Mpad, (r * S + s) * C, tiles_m);
        }
    } else {
      // stem: materialize im2col then transpose
      uint32_t Kpad = ((RSC + 7) / 8) * 8;
      auto col = at::empty({(long)M, (long)Kpad}, x.options());
      FastDiv dQ, dPQ, dC, dS;
      dQ.init(Q); dPQ.init(P * Q); dC.init(C); dS.init(S);
      size_t total = (size_t)M * Kpad;
      uint32_t blocks = std::min<size_t>((total + 255) / 256, 16384);
      im2col_kernel<<<blocks, 256, 0, cur_stream()>>>(
          (const elem_t*)x.data_ptr(), (elem_t*)col.data_ptr(), M, Kpad, RSC, C,
          H, W_, Q, S, stride, stride, pad, pad, dQ, dPQ, dC, dS);
      xt = transpose2d<elem_t>(col, Mpad).narrow(0, 0, RSC).contiguous();
    }

    // choose split-K so the grid fills the chip (~1024 blocks; 2 blocks/CU on
    // 256 CUs plus headroom for tail effects)
    uint32_t tiles_mn = ceil_div(Ko, 128) * ceil_div(RSC, 128);
    uint32_t ktiles = ceil_div(Mpad, 64);
    uint32_t splitk = std::max<uint32_t>(1, std::min<uint32_t>(ktiles, 1024 / std::max(1u, tiles_mn)));

    static const bool no_ws2 = env_flag("DTMX_DISABLE_WGRAD_WS");
    const bool ws_ok = !no_ws2 && ((size_t)Ko * RSC) % 8 == 0;
    auto dw32 = ws_ok ? wgrad_acc_ws(Ko, RSC, x)
                      : at::zeros({(long)Ko, (long)RSC},
                                  x.options().dtype(at::kFloat));
    DenseP<elem_t> pa{(const elem_t*)dyt.data_ptr(), zero_page<elem_t>(x), Ko, Mpad, Mpad};
    DenseP<elem_t> pb{(const elem_t*)xt.data_ptr(), zero_page<elem_t>(x), RSC, Mpad, Mpad};
    EpiAtomicF32<elem_t> epi{dw32.data_ptr<float>(), Ko, RSC};
    launch_gemm(pa, pb, epi, Ko, RSC, Mpad, splitk);
    // (Ko, R, S, C) fp32 -> bf16, viewed back to logical (K,C,R,S) channels_last
    if (ws_ok) {
      auto dwm = at::empty({(long)Ko, (long)R, (long)S, (long)C}, x.options());
      size_t t8 = (size_t)Ko * RSC / 8;
      cast_f32_bf16_zero_kernel<<<std::min<size_t>((t8 + 255) / 256, 2048), 256,
                                  0, cur_stream()>>>(
          dw32.data_ptr<float>(), (elem_t*)dwm.data_ptr(), t8);
      return dwm.permute({0, 3, 1, 2});
    }
    auto dw = dw32.reshape({(long)Ko, (long)R, (long)S, (long)C})
                  .to(at::kBFloat16)
                  .permute({0, 3, 1, 2});
    return dw.contiguous(at::MemoryFormat::ChannelsLast);

  });
  return at::Tensor();
}

}  // namespace dtmx
