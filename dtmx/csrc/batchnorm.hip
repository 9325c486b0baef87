// NHWC BatchNorm fwd/bwd for gfx950 (reference src/operator/nn/batch_norm.cu:
// 208-360 — redesigned for NHWC/bf16: channel reductions are V8-vectorized
// column sums with fp32 atomically-merged partials; apply passes are
// vectorized elementwise with optional fused ReLU; all hot-loop indexing is
// 32-bit with FastDiv (64-bit div/mod on the elementwise path measured ~5x).
//
// Pass structure (memory-bound):
//   fwd train: stats (x) -> finalize (tiny) -> apply (x -> y)
//   bwd:       grad-stats (x,dy[,y]) -> finalize (tiny) -> apply-dx
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "dtmx_common.h"

namespace dtmx {

static hipStream_t bn_stream() { return at::hip::getCurrentHIPStream().stream(); }

__global__ void slab_prereduce2_kernel(const float* a, const float* b,
                                       float* outa, float* outb, uint32_t C,
                                       uint32_t nslabs, uint32_t K);

// tree-reduce 8 per-thread floats across the threads sharing a channel
// vector (stride cvecs in the block), then write the block's partial row to
// a [gridDim.y][C] slab. No atomics: same-address fp32 atomicAdd serializes
// across blocks (measured ~0.2us PER BLOCK -> stats time linear in grid).
__device__ __forceinline__ void block_col_reduce(float* red, float (&v)[8],
                                                 uint32_t cv, uint32_t cpb,
                                                 uint32_t cvecs, float* slab_row) {
  const uint32_t t = threadIdx.x;
  __syncthreads();
#pragma unroll
  for (int e = 0; e < 8; ++e) red[t * 8 + e] = v[e];
  __syncthreads();
  for (uint32_t off = 128; off >= cpb; off >>= 1) {
    if (t < off) {
#pragma unroll
      for (int e = 0; e < 8; ++e) red[t * 8 + e] += red[(t + off) * 8 + e];
    }
    __syncthreads();
  }
  if (t < cpb && blockIdx.x * cpb + t < cvecs) {
#pragma unroll
    for (int e = 0; e < 8; ++e)
      slab_row[(blockIdx.x * cpb + t) * 8 + e] = red[t * 8 + e];
  }
}

// ---- forward stats: partial sum / sumsq per channel, vectorized ----------
// thread t covers channel-vector cv = t % cvecs (8 channels), row stripe
// r = r0 + t/cvecs, stepping by 256/cvecs. Requires cvecs <= 256 divisor of
// 256 (C is a power-of-two multiple of 8 in practice); general C uses the
// strided variant below.
template <typename elem_t>
__global__ void bn_stats_kernel(const elem_t* __restrict__ x, float* __restrict__ psum,
                                float* __restrict__ psumsq, uint32_t rows,
                                uint32_t cvecs, uint32_t cpb,
                                uint32_t rows_per_block) {
  using V8 = typename E8<elem_t>::v8;
  const uint32_t t = threadIdx.x;
  const uint32_t cv = blockIdx.x * cpb + t % cpb;
  const bool cv_ok = cv < cvecs;
  const uint32_t rstep = blockDim.x / cpb;
  const uint32_t r0 = blockIdx.y * rows_per_block + t / cpb;
  const uint32_t r1 = cv_ok ? min((blockIdx.y + 1) * rows_per_block, rows) : 0;
  const uint32_t C = cvecs * 8;
  float s[8] = {}, ss[8] = {};
  // 4x unrolled so four 16-B loads are in flight per wave (a single-buffer
  // loop compiles to load -> vmcnt(0) -> use and runs HBM-latency-bound:
  // measured ~20x off the bandwidth roofline).
  auto accum = [&](V8 v) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float f = (float)v[e];
      s[e] += f;
      ss[e] += f * f;
    }
  };
  uint32_t r = r0;
  for (; r + 3 * rstep < r1; r += 4 * rstep) {
    V8 v0 = *(const V8*)(x + (size_t)r * C + cv * 8);
    V8 v1 = *(const V8*)(x + (size_t)(r + rstep) * C + cv * 8);
    V8 v2 = *(const V8*)(x + (size_t)(r + 2 * rstep) * C + cv * 8);
    V8 v3 = *(const V8*)(x + (size_t)(r + 3 * rstep) * C + cv * 8);
    accum(v0); accum(v1); accum(v2); accum(v3);
  }
  for (; r < r1; r += rstep)
    accum(*(const V8*)(x + (size_t)r * C + cv * 8));
  // intra-block tree reduction, then a per-block slab row (see
  // block_col_reduce).
  __shared__ float red[256 * 8];
  block_col_reduce(red, s, cv, cpb, cvecs, psum + (size_t)blockIdx.y * C);
  block_col_reduce(red, ss, cv, cpb, cvecs, psumsq + (size_t)blockIdx.y * C);
}

// device-side body shared by the fused reduce+finalize kernels: reduces the
// two [nslabs][C] slabs for this thread's channel-vector into sa/sb (valid on
// threads t < ncv after the final barrier).
__device__ __forceinline__ void slab_reduce2_body(const float* a, const float* b,
                                                  uint32_t C, uint32_t nslabs,
                                                  uint32_t ncv,
                                                  float (&outa)[8], float (&outb)[8],
                                                  uint32_t& cv_out, uint32_t& ncv_out) {
  const uint32_t cvecs = C / 8;
  const uint32_t lanes = blockDim.x / ncv;
  const uint32_t t = threadIdx.x;
  const uint32_t cv = blockIdx.x * ncv + t % ncv;
  const uint32_t lane = t / ncv;
  cv_out = cv;
  ncv_out = ncv;
  float sa[8] = {}, sb[8] = {};
  if (cv < cvecs) {
    for (uint32_t sl = lane; sl < nslabs; sl += lanes) {
      const float* pa = a + (size_t)sl * C + cv * 8;
      const float* pb = b + (size_t)sl * C + cv * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        sa[e] += pa[e];
        sb[e] += pb[e];
      }
    }
  }
  __shared__ float red[256 * 8];
#pragma unroll
  for (int e = 0; e < 8; ++e) red[t * 8 + e] = sa[e];
  __syncthreads();
  for (uint32_t off = 128; off >= ncv; off >>= 1) {
    if (t < off) {
#pragma unroll
      for (int e = 0; e < 8; ++e) red[t * 8 + e] += red[(t + off) * 8 + e];
    }
    __syncthreads();
  }
  if (t < ncv) {
#pragma unroll
    for (int e = 0; e < 8; ++e) outa[e] = red[t * 8 + e];
  }
  __syncthreads();
#pragma unroll
  for (int e = 0; e < 8; ++e) red[t * 8 + e] = sb[e];
  __syncthreads();
  for (uint32_t off = 128; off >= ncv; off >>= 1) {
    if (t < off) {
#pragma unroll
      for (int e = 0; e < 8; ++e) red[t * 8 + e] += red[(t + off) * 8 + e];
    }
    __syncthreads();
  }
  if (t < ncv) {
#pragma unroll
    for (int e = 0; e < 8; ++e) outb[e] = red[t * 8 + e];
  }
}

// fused: slab reduce + forward finalize (mean/invstd, running stats, scale/shift)
template <typename elem_t>
__global__ void bn_reduce_finalize_kernel(
    const float* __restrict__ psum, const float* __restrict__ psumsq,
    const elem_t* __restrict__ gamma, const elem_t* __restrict__ beta,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    float* __restrict__ save_mean, float* __restrict__ save_invstd,
    float* __restrict__ scale, float* __restrict__ shift, uint32_t C,
    uint32_t nslabs, uint32_t ncv, uint32_t count, float momentum, float eps) {
  using V8 = typename E8<elem_t>::v8;
  float fsum[8], fsumsq[8];
  uint32_t cv, ncv_;
  slab_reduce2_body(psum, psumsq, C, nslabs, ncv, fsum, fsumsq, cv, ncv_);
  if (threadIdx.x < ncv_ && cv < C / 8) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      uint32_t c = cv * 8 + e;
      float mean = fsum[e] / count;
      float var = fmaxf(fsumsq[e] / count - mean * mean, 0.f);
      float invstd = rsqrtf(var + eps);
      save_mean[c] = mean;
      save_invstd[c] = invstd;
      float unbiased = count > 1 ? var * count / (count - 1) : var;
      running_mean[c] = running_mean[c] * momentum + mean * (1.f - momentum);
      running_var[c] = running_var[c] * momentum + unbiased * (1.f - momentum);
      float g = (float)gamma[c];
      scale[c] = g * invstd;
      shift[c] = (float)beta[c] - mean * g * invstd;
    }
  }
}

// fused: slab reduce + backward finalize (dgamma/dbeta + totals for dx)
template <typename elem_t>
__global__ void bn_bwd_reduce_finalize_kernel(
    const float* __restrict__ pdb, const float* __restrict__ pdg,
    elem_t* __restrict__ dgamma, elem_t* __restrict__ dbeta,
    float* __restrict__ tdb, float* __restrict__ tdg, uint32_t C,
    uint32_t nslabs, uint32_t ncv) {
  using V8 = typename E8<elem_t>::v8;
  float db[8], dg[8];
  uint32_t cv, ncv_;
  slab_reduce2_body(pdb, pdg, C, nslabs, ncv, db, dg, cv, ncv_);
  if (threadIdx.x < ncv_ && cv < C / 8) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      uint32_t c = cv * 8 + e;
      dgamma[c] = (elem_t)dg[e];
      dbeta[c] = (elem_t)db[e];
      tdb[c] = db[e];
      tdg[c] = dg[e];
    }
  }
}

// ---- parallel slab reduction: [nslabs][C] x2 -> [C] x2 -------------------
// block covers ncv channel-vectors (<=8) x (256/ncv) slab lanes; lane-strided
// accumulate then LDS tree. Replaces both the per-channel serial loop
// (measured 190us/call at nslabs=2048) and cross-block atomics.
__global__ void slab_reduce2_kernel(const float* __restrict__ a,
                                    const float* __restrict__ b,
                                    float* __restrict__ outa,
                                    float* __restrict__ outb, uint32_t C,
                                    uint32_t nslabs) {
  const uint32_t cvecs = C / 8;
  const uint32_t ncv = min(cvecs, 8u);
  const uint32_t lanes = blockDim.x / ncv;
  const uint32_t t = threadIdx.x;
  const uint32_t cv = blockIdx.x * ncv + t % ncv;
  const uint32_t lane = t / ncv;
  float sa[8] = {}, sb[8] = {};
  if (cv < cvecs) {
    for (uint32_t sl = lane; sl < nslabs; sl += lanes) {
      const float* pa = a + (size_t)sl * C + cv * 8;
      const float* pb = b + (size_t)sl * C + cv * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        sa[e] += pa[e];
        sb[e] += pb[e];
      }
    }
  }
  __shared__ float red[256 * 8];
  // tree over the slab lanes (threads sharing cv sit at stride ncv)
#pragma unroll
  for (int e = 0; e < 8; ++e) red[t * 8 + e] = sa[e];
  __syncthreads();
  for (uint32_t off = 128; off >= ncv; off >>= 1) {
    if (t < off) {
#pragma unroll
      for (int e = 0; e < 8; ++e) red[t * 8 + e] += red[(t + off) * 8 + e];
    }
    __syncthreads();
  }
  if (t < ncv && cv < cvecs) {
#pragma unroll
    for (int e = 0; e < 8; ++e) outa[cv * 8 + e] = red[t * 8 + e];
  }
  __syncthreads();
#pragma unroll
  for (int e = 0; e < 8; ++e) red[t * 8 + e] = sb[e];
  __syncthreads();
  for (uint32_t off = 128; off >= ncv; off >>= 1) {
    if (t < off) {
#pragma unroll
      for (int e = 0; e < 8; ++e) red[t * 8 + e] += red[(t + off) * 8 + e];
    }
    __syncthreads();
  }
  if (t < ncv && cv < cvecs) {
#pragma unroll
    for (int e = 0; e < 8; ++e) outb[cv * 8 + e] = red[t * 8 + e];
  }
}

// ---- finalize: mean/invstd, running stats, fused scale/shift -------------
template <typename elem_t>
__global__ void bn_finalize_kernel(const float* __restrict__ psum,
                                   const float* __restrict__ psumsq,
                                   const elem_t* __restrict__ gamma,
                                   const elem_t* __restrict__ beta,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   float* __restrict__ save_mean,
                                   float* __restrict__ save_invstd,
                                   float* __restrict__ scale,
                                   float* __restrict__ shift, uint32_t C,
                                   uint32_t nslabs, uint32_t count,
                                   float momentum, float eps) {
  using V8 = typename E8<elem_t>::v8;
  uint32_t c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float fsum = 0.f, fsumsq = 0.f;
  for (uint32_t b = 0; b < nslabs; ++b) {
    fsum += psum[(size_t)b * C + c];
    fsumsq += psumsq[(size_t)b * C + c];
  }
  float mean = fsum / count;
  float var = fmaxf(fsumsq / count - mean * mean, 0.f);
  float invstd = rsqrtf(var + eps);
  save_mean[c] = mean;
  save_invstd[c] = invstd;
  float unbiased = count > 1 ? var * count / (count - 1) : var;
  running_mean[c] = running_mean[c] * momentum + mean * (1.f - momentum);
  running_var[c] = running_var[c] * momentum + unbiased * (1.f - momentum);
  float g = (float)gamma[c];
  scale[c] = g * invstd;
  shift[c] = (float)beta[c] - mean * g * invstd;
}

template <typename elem_t>
__global__ void bn_infer_prep_kernel(const elem_t* __restrict__ gamma,
                                     const elem_t* __restrict__ beta,
                                     const float* __restrict__ running_mean,
                                     const float* __restrict__ running_var,
                                     float* __restrict__ scale,
                                     float* __restrict__ shift, uint32_t C,
                                     float eps) {
  using V8 = typename E8<elem_t>::v8;
  uint32_t c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float invstd = rsqrtf(running_var[c] + eps);
  float g = (float)gamma[c];
  scale[c] = g * invstd;
  shift[c] = (float)beta[c] - running_mean[c] * g * invstd;
}

// ---- apply: y = x*scale + shift (+relu), V8, 32-bit indexing ---------
template <typename elem_t>
__global__ void bn_apply_kernel(const elem_t* __restrict__ x, elem_t* __restrict__ y,
                                const float* __restrict__ scale,
                                const float* __restrict__ shift,
                                const elem_t* __restrict__ residual,  // nullable
                                uint32_t total8, FastDiv dcv, int relu) {
  using V8 = typename E8<elem_t>::v8;
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  const uint32_t stride = gridDim.x * blockDim.x;
  auto body = [&](uint32_t ii, V8 v, V8 res) {
    uint32_t q = dcv.div(ii);
    uint32_t c0 = dcv.mod(ii, q) * 8;
    V8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float r = (float)v[e] * scale[c0 + e] + shift[c0 + e];
      if (residual) r += (float)res[e];
      if (relu) r = fmaxf(r, 0.f);
      o[e] = (elem_t)r;
    }
    *(V8*)(y + (size_t)ii * 8) = o;
  };
  V8 zed = {};
  // 2x unrolled so 2-4 16-B loads are in flight per wave (single-chunk loop
  // compiles to load -> wait -> use: HBM-latency-bound)
  for (; i + stride < total8; i += 2 * stride) {
    V8 v0 = *(const V8*)(x + (size_t)i * 8);
    V8 v1 = *(const V8*)(x + (size_t)(i + stride) * 8);
    V8 r0 = residual ? *(const V8*)(residual + (size_t)i * 8) : zed;
    V8 r1 = residual ? *(const V8*)(residual + (size_t)(i + stride) * 8) : zed;
    body(i, v0, r0);
    body(i + stride, v1, r1);
  }
  for (; i < total8; i += stride) {
    V8 v = *(const V8*)(x + (size_t)i * 8);
    V8 res = residual ? *(const V8*)(residual + (size_t)i * 8) : zed;
    body(i, v, res);
  }
}

// ---- backward stats: per-channel sum(dy), sum(dy*xhat), vectorized -------
template <typename elem_t>
__global__ void bn_bwd_stats_kernel(const elem_t* __restrict__ x,
                                    const elem_t* __restrict__ dy,
                                    const elem_t* __restrict__ y,  // relu mask
                                    const float* __restrict__ save_mean,
                                    const float* __restrict__ save_invstd,
                                    float* __restrict__ pdb, float* __restrict__ pdg,
                                    uint32_t rows, uint32_t cvecs, uint32_t cpb,
                                    uint32_t rows_per_block, int relu) {
  using V8 = typename E8<elem_t>::v8;
  const uint32_t t = threadIdx.x;
  const uint32_t cv = blockIdx.x * cpb + t % cpb;
  const bool cv_ok = cv < cvecs;
  const uint32_t rstep = blockDim.x / cpb;
  const uint32_t r0 = blockIdx.y * rows_per_block + t / cpb;
  const uint32_t r1 = cv_ok ? min((blockIdx.y + 1) * rows_per_block, rows) : 0;
  const uint32_t C = cvecs * 8;
  float mean[8], invstd[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    mean[e] = cv_ok ? save_mean[cv * 8 + e] : 0.f;
    invstd[e] = cv_ok ? save_invstd[cv * 8 + e] : 0.f;
  }
  float db[8] = {}, dg[8] = {};
  auto accum = [&](V8 xv, V8 gv, V8 yv) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float g = (float)gv[e];
      if (relu && (float)yv[e] <= 0.f) g = 0.f;
      db[e] += g;
      dg[e] += g * ((float)xv[e] - mean[e]) * invstd[e];
    }
  };
  V8 zed = {};
  uint32_t r = r0;
  for (; r + rstep < r1; r += 2 * rstep) {  // 2x unroll: 4-6 loads in flight
    size_t o0 = (size_t)r * C + cv * 8, o1 = (size_t)(r + rstep) * C + cv * 8;
    V8 x0 = *(const V8*)(x + o0), x1 = *(const V8*)(x + o1);
    V8 g0 = *(const V8*)(dy + o0), g1 = *(const V8*)(dy + o1);
    V8 y0 = relu ? *(const V8*)(y + o0) : zed;
    V8 y1 = relu ? *(const V8*)(y + o1) : zed;
    accum(x0, g0, y0);
    accum(x1, g1, y1);
  }
  for (; r < r1; r += rstep) {
    size_t o0 = (size_t)r * C + cv * 8;
    accum(*(const V8*)(x + o0), *(const V8*)(dy + o0),
          relu ? *(const V8*)(y + o0) : zed);
  }
  __shared__ float red[256 * 8];
  block_col_reduce(red, db, cv, cpb, cvecs, pdb + (size_t)blockIdx.y * C);
  block_col_reduce(red, dg, cv, cpb, cvecs, pdg + (size_t)blockIdx.y * C);
}

template <typename elem_t>
__global__ void bn_bwd_finalize_kernel(const float* __restrict__ pdb,
                                       const float* __restrict__ pdg,
                                       elem_t* __restrict__ dgamma,
                                       elem_t* __restrict__ dbeta,
                                       float* __restrict__ tdb,
                                       float* __restrict__ tdg, uint32_t C,
                                       uint32_t nslabs) {
  using V8 = typename E8<elem_t>::v8;
  uint32_t c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float db = 0.f, dg = 0.f;
  for (uint32_t b = 0; b < nslabs; ++b) {
    db += pdb[(size_t)b * C + c];
    dg += pdg[(size_t)b * C + c];
  }
  dgamma[c] = (elem_t)dg;
  dbeta[c] = (elem_t)db;
  tdb[c] = db;
  tdg[c] = dg;
}

// dx = gamma*invstd * (dy - dbeta/M - xhat * dgamma/M)
template <typename elem_t>
__global__ void bn_bwd_dx_kernel(const elem_t* __restrict__ x,
                                 const elem_t* __restrict__ dy,
                                 const elem_t* __restrict__ y,
                                 const float* __restrict__ save_mean,
                                 const float* __restrict__ save_invstd,
                                 const elem_t* __restrict__ gamma,
                                 const float* __restrict__ pdb,
                                 const float* __restrict__ pdg,
                                 elem_t* __restrict__ dx,
                                 elem_t* __restrict__ dres,  // nullable
                                 uint32_t total8, FastDiv dcv, float inv_count,
                                 int relu) {
  using V8 = typename E8<elem_t>::v8;
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  const uint32_t stride = gridDim.x * blockDim.x;
  auto body = [&](uint32_t ii, V8 xv, V8 gv, V8 yv) {
    uint32_t q = dcv.div(ii);
    uint32_t c0 = dcv.mod(ii, q) * 8;
    size_t off = (size_t)ii * 8;
    V8 o, om;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      uint32_t c = c0 + e;
      float g = (float)gv[e];
      if (relu && (float)yv[e] <= 0.f) g = 0.f;
      if (dres) om[e] = (elem_t)g;
      float invstd = save_invstd[c];
      float xh = ((float)xv[e] - save_mean[c]) * invstd;
      o[e] = (elem_t)((float)gamma[c] * invstd *
                      (g - pdb[c] * inv_count - xh * pdg[c] * inv_count));
    }
    *(V8*)(dx + off) = o;
    if (dres) *(V8*)(dres + off) = om;
  };
  V8 zed = {};
  // 2x unrolled: 4-6 loads in flight per wave instead of load->wait->use
  for (; i + stride < total8; i += 2 * stride) {
    size_t o0 = (size_t)i * 8, o1 = (size_t)(i + stride) * 8;
    V8 x0 = *(const V8*)(x + o0), x1 = *(const V8*)(x + o1);
    V8 g0 = *(const V8*)(dy + o0), g1 = *(const V8*)(dy + o1);
    V8 y0 = relu ? *(const V8*)(y + o0) : zed;
    V8 y1 = relu ? *(const V8*)(y + o1) : zed;
    body(i, x0, g0, y0);
    body(i + stride, x1, g1, y1);
  }
  for (; i < total8; i += stride) {
    size_t o0 = (size_t)i * 8;
    body(i, *(const V8*)(x + o0), *(const V8*)(dy + o0),
         relu ? *(const V8*)(y + o0) : zed);
  }
}

// ---- coefficient-table dx: dx = ka[c]*dy + kc[c]*x + kb[c] ---------------
// The closed form above folds gamma/invstd/mean/tdb/tdg into three
// per-channel tables (ka = gamma*invstd, kc = -ka*invstd*tdg/M,
// kb = -ka*tdb/M - kc*mean), cutting the per-chunk table traffic from five
// arrays (9 dwordx4) to three (6) and halving the per-element FLOPs.
// DTMX_BN_DX_COEF=0 restores the direct form.
template <typename elem_t>
__global__ void bn_dx_coef_kernel(const float* __restrict__ tdb,
                                  const float* __restrict__ tdg,
                                  const elem_t* __restrict__ gamma,
                                  const float* __restrict__ save_mean,
                                  const float* __restrict__ save_invstd,
                                  float* __restrict__ ka, float* __restrict__ kb,
                                  float* __restrict__ kc, uint32_t C,
                                  float inv_count) {
  uint32_t c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float invstd = save_invstd[c];
  float a = (float)gamma[c] * invstd;
  float cc = -a * invstd * tdg[c] * inv_count;
  ka[c] = a;
  kc[c] = cc;
  kb[c] = -a * tdb[c] * inv_count - cc * save_mean[c];
}

template <typename elem_t>
__global__ void bn_bwd_dx_coef_kernel(const elem_t* __restrict__ x,
                                      const elem_t* __restrict__ dy,
                                      const elem_t* __restrict__ y,
                                      const float* __restrict__ ka,
                                      const float* __restrict__ kb,
                                      const float* __restrict__ kc,
                                      elem_t* __restrict__ dx,
                                      elem_t* __restrict__ dres,  // nullable
                                      uint32_t total8, FastDiv dcv, int relu) {
  using V8 = typename E8<elem_t>::v8;
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  const uint32_t stride = gridDim.x * blockDim.x;
  auto body = [&](uint32_t ii, V8 xv, V8 gv, V8 yv) {
    uint32_t q = dcv.div(ii);
    uint32_t c0 = dcv.mod(ii, q) * 8;
    size_t off = (size_t)ii * 8;
    V8 o, om;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      uint32_t c = c0 + e;
      float g = (float)gv[e];
      if (relu && (float)yv[e] <= 0.f) g = 0.f;
      if (dres) om[e] = (elem_t)g;
      o[e] = (elem_t)(ka[c] * g + kc[c] * (float)xv[e] + kb[c]);
    }
    *(V8*)(dx + off) = o;
    if (dres) *(V8*)(dres + off) = om;
  };
  V8 zed = {};
  for (; i + stride < total8; i += 2 * stride) {
    size_t o0 = (size_t)i * 8, o1 = (size_t)(i + stride) * 8;
    V8 x0 = *(const V8*)(x + o0), x1 = *(const V8*)(x + o1);
    V8 g0 = *(const V8*)(dy + o0), g1 = *(const V8*)(dy + o1);
    V8 y0 = relu ? *(const V8*)(y + o0) : zed;
    V8 y1 = relu ? *(const V8*)(y + o1) : zed;
    body(i, x0, g0, y0);
    body(i + stride, x1, g1, y1);
  }
  for (; i < total8; i += stride) {
    size_t o0 = (size_t)i * 8;
    body(i, *(const V8*)(x + o0), *(const V8*)(dy + o0),
         relu ? *(const V8*)(y + o0) : zed);
  }
}

// ---- fixed-column variants: thread owns ONE channel-vector column --------
// The strided-chunk kernels re-load the per-channel tables (and run a
// FastDiv) for every 16-B chunk; torch's pure d=a+b triad reaches 6.06 TB/s
// on this box where bn_bwd_dx measured 4.87 — the table traffic is the gap.
// Owning a fixed cv per thread hoists the tables into registers before the
// row loop; coalescing stays intact (cpb consecutive lanes cover cpb*16 B
// of one row). Geometry = bn_stats_kernel's (cv, strided rows).
template <typename elem_t>
__global__ void bn_apply_col_kernel(const elem_t* __restrict__ x,
                                    elem_t* __restrict__ y,
                                    const float* __restrict__ scale,
                                    const float* __restrict__ shift,
                                    const elem_t* __restrict__ residual,
                                    uint32_t rows, uint32_t cvecs, uint32_t cpb,
                                    uint32_t rows_per_block, int relu) {
  using V8 = typename E8<elem_t>::v8;
  const uint32_t t = threadIdx.x;
  const uint32_t cv = blockIdx.x * cpb + t % cpb;
  if (cv >= cvecs) return;
  const uint32_t rstep = blockDim.x / cpb;
  const uint32_t r0 = blockIdx.y * rows_per_block + t / cpb;
  const uint32_t r1 = min((blockIdx.y + 1) * rows_per_block, rows);
  float sc[8], sh[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    sc[e] = scale[cv * 8 + e];
    sh[e] = shift[cv * 8 + e];
  }
  auto body = [&](size_t off, V8 v, V8 res) {
    V8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float r = (float)v[e] * sc[e] + sh[e];
      if (residual) r += (float)res[e];
      if (relu) r = fmaxf(r, 0.f);
      o[e] = (elem_t)r;
    }
    *(V8*)(y + off) = o;
  };
  V8 zed = {};
  uint32_t r = r0;
  for (; r + rstep < r1; r += 2 * rstep) {
    size_t o0 = ((size_t)r * cvecs + cv) * 8;
    size_t o1 = ((size_t)(r + rstep) * cvecs + cv) * 8;
    V8 v0 = *(const V8*)(x + o0), v1 = *(const V8*)(x + o1);
    V8 r0v = residual ? *(const V8*)(residual + o0) : zed;
    V8 r1v = residual ? *(const V8*)(residual + o1) : zed;
    body(o0, v0, r0v);
    body(o1, v1, r1v);
  }
  for (; r < r1; r += rstep) {
    size_t o0 = ((size_t)r * cvecs + cv) * 8;
    body(o0, *(const V8*)(x + o0),
         residual ? *(const V8*)(residual + o0) : zed);
  }
}

template <typename elem_t>
__global__ void bn_bwd_dx_col_kernel(const elem_t* __restrict__ x,
                                     const elem_t* __restrict__ dy,
                                     const elem_t* __restrict__ y,
                                     const float* __restrict__ ka,
                                     const float* __restrict__ kb,
                                     const float* __restrict__ kc,
                                     elem_t* __restrict__ dx,
                                     elem_t* __restrict__ dres,
                                     uint32_t rows, uint32_t cvecs,
                                     uint32_t cpb, uint32_t rows_per_block,
                                     int relu) {
  using V8 = typename E8<elem_t>::v8;
  const uint32_t t = threadIdx.x;
  const uint32_t cv = blockIdx.x * cpb + t % cpb;
  if (cv >= cvecs) return;
  const uint32_t rstep = blockDim.x / cpb;
  const uint32_t r0 = blockIdx.y * rows_per_block + t / cpb;
  const uint32_t r1 = min((blockIdx.y + 1) * rows_per_block, rows);
  float a8[8], b8[8], c8[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    a8[e] = ka[cv * 8 + e];
    b8[e] = kb[cv * 8 + e];
    c8[e] = kc[cv * 8 + e];
  }
  auto body = [&](size_t off, V8 xv, V8 gv, V8 yv) {
    V8 o, om;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float g = (float)gv[e];
      if (relu && (float)yv[e] <= 0.f) g = 0.f;
      if (dres) om[e] = (elem_t)g;
      o[e] = (elem_t)(a8[e] * g + c8[e] * (float)xv[e] + b8[e]);
    }
    *(V8*)(dx + off) = o;
    if (dres) *(V8*)(dres + off) = om;
  };
  V8 zed = {};
  uint32_t r = r0;
  for (; r + rstep < r1; r += 2 * rstep) {
    size_t o0 = ((size_t)r * cvecs + cv) * 8;
    size_t o1 = ((size_t)(r + rstep) * cvecs + cv) * 8;
    V8 x0 = *(const V8*)(x + o0), x1 = *(const V8*)(x + o1);
    V8 g0 = *(const V8*)(dy + o0), g1 = *(const V8*)(dy + o1);
    V8 y0 = relu ? *(const V8*)(y + o0) : zed;
    V8 y1 = relu ? *(const V8*)(y + o1) : zed;
    body(o0, x0, g0, y0);
    body(o1, x1, g1, y1);
  }
  for (; r < r1; r += rstep) {
    size_t o0 = ((size_t)r * cvecs + cv) * 8;
    body(o0, *(const V8*)(x + o0), *(const V8*)(dy + o0),
         relu ? *(const V8*)(y + o0) : zed);
  }
}

static bool bn_col_on() {
  static const bool on = [] {
    const char* e = getenv("DTMX_BN_COL");
    return !e || e[0] != '0';  // default ON (A/B'd on-box)
  }();
  return on;
}

static bool bn_dx_coef_on() {
  static const bool on = [] {
    const char* e = getenv("DTMX_BN_DX_COEF");
    return !e || e[0] != '0';  // default ON (A/B'd on-box)
  }();
  return on;
}

// ============================================================== host side ==

// stats geometry: cpb = channel-vectors per block, the largest power-of-2
// divisor of cvecs (<=256) — exact thread utilization for any C % 8 == 0
// (e.g. inception's 80/768/1280-channel BNs: cvecs 10 -> cpb 2, grid.x 5).
static uint32_t bn_cpb(uint32_t cvecs) {
  uint32_t cpb = cvecs & (~cvecs + 1);  // lowest set bit = largest pow2 divisor
  while (cpb < cvecs && cpb * 2 <= 256 && cvecs % (cpb * 2) == 0) cpb *= 2;
  return std::min<uint32_t>(cpb, 256);
}

static void bn_grid(uint32_t rows, uint32_t cvecs, uint32_t cpb, dim3& grid,
                    uint32_t& rows_per_block) {
  uint32_t rstep = 256 / cpb;
  // >=16 strip iterations per thread so the per-block reduce tail amortizes;
  // cap at 2048 blocks (8/CU) for latency hiding + BW saturation.
  uint32_t rb = std::min<uint32_t>(2048, std::max<uint32_t>(1, rows / (rstep * 16)));
  rows_per_block = (rows + rb - 1) / rb;
  rb = (rows + rows_per_block - 1) / rows_per_block;
  grid = dim3((cvecs + cpb - 1) / cpb, rb);
}

// dispatch helper: fixed-column kernel (tables hoisted to registers) by
// default, strided-chunk kernel under DTMX_BN_COL=0
template <typename elem_t>
static void launch_bn_apply(const elem_t* x, elem_t* y, const float* scale,
                            const float* shift, const elem_t* residual,
                            uint32_t rows, uint32_t cvecs, int relu,
                            hipStream_t s) {
  if (bn_col_on()) {
    uint32_t cpb = bn_cpb(cvecs);
    dim3 grid;
    uint32_t rpb;
    bn_grid(rows, cvecs, cpb, grid, rpb);
    bn_apply_col_kernel<<<grid, 256, 0, s>>>(x, y, scale, shift, residual,
                                             rows, cvecs, cpb, rpb, relu);
  } else {
    uint32_t total8 = rows * cvecs;
    FastDiv dcv;
    dcv.init(cvecs);
    uint32_t blocks = std::min<uint32_t>((total8 + 255) / 256, 2048);
    bn_apply_kernel<<<blocks, 256, 0, s>>>(x, y, scale, shift, residual,
                                           total8, dcv, relu);
  }
}

template <typename elem_t>
static void launch_bn_dx_coef(const elem_t* x, const elem_t* dy,
                              const elem_t* y, const float* ka,
                              const float* kb, const float* kc, elem_t* dx,
                              elem_t* dres, uint32_t rows, uint32_t cvecs,
                              int relu, hipStream_t s) {
  if (bn_col_on()) {
    uint32_t cpb = bn_cpb(cvecs);
    dim3 grid;
    uint32_t rpb;
    bn_grid(rows, cvecs, cpb, grid, rpb);
    bn_bwd_dx_col_kernel<<<grid, 256, 0, s>>>(x, dy, y, ka, kb, kc, dx, dres,
                                              rows, cvecs, cpb, rpb, relu);
  } else {
    uint32_t total8 = rows * cvecs;
    FastDiv dcv;
    dcv.init(cvecs);
    uint32_t blocks = std::min<uint32_t>((total8 + 255) / 256, 2048);
    bn_bwd_dx_coef_kernel<<<blocks, 256, 0, s>>>(x, dy, y, ka, kb, kc, dx,
                                                 dres, total8, dcv, relu);
  }
}

std::vector<at::Tensor> bn_fwd_train(const at::Tensor& x, const at::Tensor& gamma,
                                     const at::Tensor& beta, at::Tensor running_mean,
                                     at::Tensor running_var, double momentum,
                                     double eps, bool fuse_relu,
                                     const c10::optional<at::Tensor>& residual,
                                     const c10::optional<at::Tensor>& pre_psum,
                                     const c10::optional<at::Tensor>& pre_psumsq) {
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "bn: x must be NHWC");
  uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C % 8 == 0, "bn: C must be a multiple of 8, got ", C);
  uint32_t rows = N * H * W, cvecs = C / 8;
  uint32_t cpb = bn_cpb(cvecs);
  auto opt_f = x.options().dtype(at::kFloat);
  dim3 grid;
  uint32_t rpb;
  bn_grid(rows, cvecs, cpb, grid, rpb);
  uint32_t nslabs = grid.y;
  at::Tensor psum, psumsq;
  const bool have_pre =
      pre_psum.has_value() && pre_psum->defined() && pre_psum->numel() > 0;
  if (have_pre) {
    // fused path: the producing conv's epilogue already wrote the slabs
    psum = *pre_psum;
    psumsq = *pre_psumsq;
    nslabs = psum.size(0);
    if (nslabs > 1024) {  // epilogue slabs have tiles_m rows: pre-reduce
      uint32_t K = 512;
      auto pa = at::empty({(long)K, (long)C}, opt_f);
      auto pb = at::empty({(long)K, (long)C}, opt_f);
      uint32_t total = K * C;
      slab_prereduce2_kernel<<<(total + 255) / 256, 256, 0, bn_stream()>>>(
          psum.data_ptr<float>(), psumsq.data_ptr<float>(),
          pa.data_ptr<float>(), pb.data_ptr<float>(), C, nslabs, K);
      psum = pa;
      psumsq = pb;
      nslabs = K;
    }
  } else {
    psum = at::empty({(long)nslabs, (long)C}, opt_f);
    psumsq = at::empty({(long)nslabs, (long)C}, opt_f);
  }
  auto save_mean = at::empty({(long)C}, opt_f), save_invstd = at::empty({(long)C}, opt_f);
  auto scale = at::empty({(long)C}, opt_f), shift = at::empty({(long)C}, opt_f);
  auto y = at::empty_like(x);
  auto s = bn_stream();
  DTMX_DISPATCH_16(x.scalar_type(), "bn_fwd", {
    if (!have_pre)
      bn_stats_kernel<<<grid, 256, 0, s>>>((const elem_t*)x.data_ptr(),
                                           psum.data_ptr<float>(),
                                           psumsq.data_ptr<float>(), rows, cvecs,
                                           cpb, rpb);
    uint32_t ncv = std::min(cpb, 8u);
    bn_reduce_finalize_kernel<<<(cvecs + ncv - 1) / ncv, 256, 0, s>>>(
        psum.data_ptr<float>(), psumsq.data_ptr<float>(),
        (const elem_t*)gamma.data_ptr(), (const elem_t*)beta.data_ptr(),
        running_mean.data_ptr<float>(), running_var.data_ptr<float>(),
        save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),
        scale.data_ptr<float>(), shift.data_ptr<float>(), C, nslabs, ncv, rows,
        momentum, eps);
    launch_bn_apply<elem_t>(
        (const elem_t*)x.data_ptr(), (elem_t*)y.data_ptr(),
        scale.data_ptr<float>(), shift.data_ptr<float>(),
        residual.has_value() ? (const elem_t*)residual->data_ptr() : nullptr,
        rows, cvecs, fuse_relu ? 1 : 0, s);
  });
  return {y, save_mean, save_invstd};
}

at::Tensor bn_fwd_infer(const at::Tensor& x, const at::Tensor& gamma,
                        const at::Tensor& beta, const at::Tensor& running_mean,
                        const at::Tensor& running_var, double eps, bool fuse_relu,
                        const c10::optional<at::Tensor>& residual) {
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "bn: x must be NHWC");
  uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C % 8 == 0, "bn: C must be a multiple of 8");
  uint32_t rows = N * H * W, cvecs = C / 8;
  auto opt_f = x.options().dtype(at::kFloat);
  auto scale = at::empty({(long)C}, opt_f), shift = at::empty({(long)C}, opt_f);
  auto y = at::empty_like(x);
  auto s = bn_stream();
  DTMX_DISPATCH_16(x.scalar_type(), "bn_infer", {
    bn_infer_prep_kernel<<<(C + 255) / 256, 256, 0, s>>>(
        (const elem_t*)gamma.data_ptr(), (const elem_t*)beta.data_ptr(),
        running_mean.data_ptr<float>(), running_var.data_ptr<float>(),
        scale.data_ptr<float>(), shift.data_ptr<float>(), C, eps);
    launch_bn_apply<elem_t>(
        (const elem_t*)x.data_ptr(), (elem_t*)y.data_ptr(),
        scale.data_ptr<float>(), shift.data_ptr<float>(),
        residual.has_value() ? (const elem_t*)residual->data_ptr() : nullptr,
        rows, cvecs, fuse_relu ? 1 : 0, s);
  });
  return y;
}

std::vector<at::Tensor> bn_bwd(const at::Tensor& x, const at::Tensor& dy,
                               const at::Tensor& gamma, const at::Tensor& save_mean,
                               const at::Tensor& save_invstd, bool fuse_relu,
                               const at::Tensor& y, bool want_dres) {
  uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C % 8 == 0, "bn: C must be a multiple of 8, got ", C);
  uint32_t rows = N * H * W, cvecs = C / 8;
  uint32_t cpb = bn_cpb(cvecs);
  auto opt_f = x.options().dtype(at::kFloat);
  dim3 grid;
  uint32_t rpb;
  bn_grid(rows, cvecs, cpb, grid, rpb);
  uint32_t nslabs = grid.y;
  auto pdb = at::empty({(long)nslabs, (long)C}, opt_f);
  auto pdg = at::empty({(long)nslabs, (long)C}, opt_f);
  auto tdb = at::empty({(long)C}, opt_f), tdg = at::empty({(long)C}, opt_f);
  auto dgamma = at::empty({(long)C}, x.options());
  auto dbeta = at::empty({(long)C}, x.options());
  auto dx = at::empty_like(x);
  at::Tensor dres;
  if (want_dres) dres = at::empty_like(x);
  auto s = bn_stream();
  DTMX_DISPATCH_16(x.scalar_type(), "bn_bwd", {
    bn_bwd_stats_kernel<<<grid, 256, 0, s>>>(
        (const elem_t*)x.data_ptr(), (const elem_t*)dy.data_ptr(),
        (const elem_t*)y.data_ptr(), save_mean.data_ptr<float>(),
        save_invstd.data_ptr<float>(), pdb.data_ptr<float>(),
        pdg.data_ptr<float>(), rows, cvecs, cpb, rpb, fuse_relu ? 1 : 0);
    uint32_t ncv = std::min(cpb, 8u);
    bn_bwd_reduce_finalize_kernel<<<(cvecs + ncv - 1) / ncv, 256, 0, s>>>(
        pdb.data_ptr<float>(), pdg.data_ptr<float>(), (elem_t*)dgamma.data_ptr(),
        (elem_t*)dbeta.data_ptr(), tdb.data_ptr<float>(), tdg.data_ptr<float>(),
        C, nslabs, ncv);
    uint32_t total8 = rows * cvecs;
    FastDiv dcv;
    dcv.init(cvecs);
    uint32_t blocks = std::min<uint32_t>((total8 + 255) / 256, 2048);
    if (bn_dx_coef_on()) {
      auto ka = at::empty({(long)C}, opt_f), kb = at::empty({(long)C}, opt_f),
           kc = at::empty({(long)C}, opt_f);
      bn_dx_coef_kernel<<<(C + 255) / 256, 256, 0, s>>>(
          tdb.data_ptr<float>(), tdg.data_ptr<float>(),
          (const elem_t*)gamma.data_ptr(), save_mean.data_ptr<float>(),
          save_invstd.data_ptr<float>(), ka.data_ptr<float>(),
          kb.data_ptr<float>(), kc.data_ptr<float>(), C, 1.f / rows);
      launch_bn_dx_coef<elem_t>(
          (const elem_t*)x.data_ptr(), (const elem_t*)dy.data_ptr(),
          (const elem_t*)y.data_ptr(), ka.data_ptr<float>(),
          kb.data_ptr<float>(), kc.data_ptr<float>(), (elem_t*)dx.data_ptr(),
          want_dres ? (elem_t*)dres.data_ptr() : nullptr, rows, cvecs,
          fuse_relu ? 1 : 0, s);
    } else {
      bn_bwd_dx_kernel<<<blocks, 256, 0, s>>>(
          (const elem_t*)x.data_ptr(), (const elem_t*)dy.data_ptr(),
          (const elem_t*)y.data_ptr(), save_mean.data_ptr<float>(),
          save_invstd.data_ptr<float>(), (const elem_t*)gamma.data_ptr(),
          tdb.data_ptr<float>(), tdg.data_ptr<float>(), (elem_t*)dx.data_ptr(),
          want_dres ? (elem_t*)dres.data_ptr() : nullptr, total8, dcv,
          1.f / rows, fuse_relu ? 1 : 0);
    }
  });
  if (want_dres) return {dx, dgamma, dbeta, dres};
  return {dx, dgamma, dbeta};
}

// Wide-grid pre-reduction for very tall slab stacks: [nslabs][C] x2 ->
// [K][C] x2 with out[k][c] = sum_{sl ≡ k (mod K)} in[sl][c]. The EpiBnBwd
// epilogue emits one slab row per 128-row GEMM tile — up to ~25k rows at
// bs1024 — and the channel-parallel finalize kernel launches only C/64
// blocks, so reducing the stack there serializes on a handful of CUs
// (measured 5.6 ms/step). Here thread (k,c) owns a strided column: K*C
// threads fill the chip and every warp read is coalesced in c.
__global__ void slab_prereduce2_kernel(const float* __restrict__ a,
                                       const float* __restrict__ b,
                                       float* __restrict__ outa,
                                       float* __restrict__ outb, uint32_t C,
                                       uint32_t nslabs, uint32_t K) {
  uint32_t tid = blockIdx.x * blockDim.x + threadIdx.x;
  if (tid >= K * C) return;
  uint32_t k = tid / C, c = tid % C;
  float sa = 0.f, sb = 0.f;
  for (uint32_t sl = k; sl < nslabs; sl += K) {
    sa += a[(size_t)sl * C + c];
    sb += b[(size_t)sl * C + c];
  }
  outa[(size_t)k * C + c] = sa;
  outb[(size_t)k * C + c] = sb;
}

// Finalize the [tiles_m][C] partial-sum slabs produced by the EpiBnBwd
// GEMM epilogue (gemm_conv.hip conv_dgrad_bnfuse): one reduce+finalize
// launch -> {dgamma, dbeta, tdb, tdg}. `like` supplies the elem dtype.
std::vector<at::Tensor> bn_bwd_finalize_slabs(const at::Tensor& pdb,
                                              const at::Tensor& pdg,
                                              const at::Tensor& like) {
  uint32_t nslabs = pdb.size(0), C = pdb.size(1);
  TORCH_CHECK(C % 8 == 0, "bn: C must be a multiple of 8, got ", C);
  uint32_t cvecs = C / 8;
  auto opt_f = pdb.options();
  auto tdb = at::empty({(long)C}, opt_f), tdg = at::empty({(long)C}, opt_f);
  auto dgamma = at::empty({(long)C}, like.options());
  auto dbeta = at::empty({(long)C}, like.options());
  auto s = bn_stream();
  const float* pa = pdb.data_ptr<float>();
  const float* pb = pdg.data_ptr<float>();
  at::Tensor preda, predb;
  if (nslabs > 1024) {  // pre-reduce tall stacks across the whole chip
    uint32_t K = 512;
    preda = at::empty({(long)K, (long)C}, opt_f);
    predb = at::empty({(long)K, (long)C}, opt_f);
    uint32_t total = K * C;
    slab_prereduce2_kernel<<<(total + 255) / 256, 256, 0, s>>>(
        pa, pb, preda.data_ptr<float>(), predb.data_ptr<float>(), C, nslabs, K);
    pa = preda.data_ptr<float>();
    pb = predb.data_ptr<float>();
    nslabs = K;
  }
  DTMX_DISPATCH_16(like.scalar_type(), "bn_bwd_finalize_slabs", {
    uint32_t ncv = std::min(bn_cpb(cvecs), 8u);
    bn_bwd_reduce_finalize_kernel<<<(cvecs + ncv - 1) / ncv, 256, 0, s>>>(
        pa, pb, (elem_t*)dgamma.data_ptr(),
        (elem_t*)dbeta.data_ptr(), tdb.data_ptr<float>(), tdg.data_ptr<float>(),
        C, nslabs, ncv);
  });
  return {dgamma, dbeta, tdb, tdg};
}

// ---- SyncBatchNorm entries (reference contrib/sync_batch_norm.cu) --------
// Cross-rank BN = local per-channel sums -> RCCL all-reduce (Python side) ->
// finalize/apply with the GLOBAL count. These entries split bn_fwd_train /
// bn_bwd at exactly that seam, reusing the same kernels.

std::vector<at::Tensor> bn_local_sums(const at::Tensor& x) {
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "bn: x must be NHWC");
  uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C % 8 == 0, "bn: C must be a multiple of 8, got ", C);
  uint32_t rows = N * H * W, cvecs = C / 8;
  uint32_t cpb = bn_cpb(cvecs);
  auto opt_f = x.options().dtype(at::kFloat);
  dim3 grid;
  uint32_t rpb;
  bn_grid(rows, cvecs, cpb, grid, rpb);
  uint32_t nslabs = grid.y;
  auto psum = at::empty({(long)nslabs, (long)C}, opt_f);
  auto psumsq = at::empty({(long)nslabs, (long)C}, opt_f);
  auto sum = at::empty({(long)C}, opt_f), sumsq = at::empty({(long)C}, opt_f);
  auto s = bn_stream();
  DTMX_DISPATCH_16(x.scalar_type(), "bn_local_sums", {
    bn_stats_kernel<<<grid, 256, 0, s>>>((const elem_t*)x.data_ptr(),
                                         psum.data_ptr<float>(),
                                         psumsq.data_ptr<float>(), rows, cvecs,
                                         cpb, rpb);
  });
  uint32_t ncv = std::min(cvecs, 8u);
  slab_reduce2_kernel<<<(cvecs + ncv - 1) / ncv, 256, 0, s>>>(
      psum.data_ptr<float>(), psumsq.data_ptr<float>(), sum.data_ptr<float>(),
      sumsq.data_ptr<float>(), C, nslabs);
  return {sum, sumsq};
}

std::vector<at::Tensor> bn_fwd_presummed(
    const at::Tensor& x, const at::Tensor& gamma, const at::Tensor& beta,
    at::Tensor running_mean, at::Tensor running_var, double momentum, double eps,
    bool fuse_relu, const c10::optional<at::Tensor>& residual,
    const at::Tensor& sum, const at::Tensor& sumsq, long count) {
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "bn: x must be NHWC");
  uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  uint32_t rows = N * H * W, cvecs = C / 8;
  auto opt_f = x.options().dtype(at::kFloat);
  auto save_mean = at::empty({(long)C}, opt_f), save_invstd = at::empty({(long)C}, opt_f);
  auto scale = at::empty({(long)C}, opt_f), shift = at::empty({(long)C}, opt_f);
  auto y = at::empty_like(x);
  auto s = bn_stream();
  (void)rows;
  DTMX_DISPATCH_16(x.scalar_type(), "bn_fwd_presummed", {
    bn_finalize_kernel<<<(C + 255) / 256, 256, 0, s>>>(
        sum.data_ptr<float>(), sumsq.data_ptr<float>(),
        (const elem_t*)gamma.data_ptr(), (const elem_t*)beta.data_ptr(),
        running_mean.data_ptr<float>(), running_var.data_ptr<float>(),
        save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),
        scale.data_ptr<float>(), shift.data_ptr<float>(), C, /*nslabs=*/1,
        (uint32_t)count, momentum, eps);
    launch_bn_apply<elem_t>(
        (const elem_t*)x.data_ptr(), (elem_t*)y.data_ptr(),
        scale.data_ptr<float>(), shift.data_ptr<float>(),
        residual.has_value() ? (const elem_t*)residual->data_ptr() : nullptr,
        N * H * W, cvecs, fuse_relu ? 1 : 0, s);
  });
  return {y, save_mean, save_invstd};
}

std::vector<at::Tensor> bn_bwd_sums(const at::Tensor& x, const at::Tensor& dy,
                                    const at::Tensor& y,
                                    const at::Tensor& save_mean,
                                    const at::Tensor& save_invstd, bool fuse_relu) {
  uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  uint32_t rows = N * H * W, cvecs = C / 8;
  uint32_t cpb = bn_cpb(cvecs);
  auto opt_f = x.options().dtype(at::kFloat);
  dim3 grid;
  uint32_t rpb;
  bn_grid(rows, cvecs, cpb, grid, rpb);
  uint32_t nslabs = grid.y;
  auto pdb = at::empty({(long)nslabs, (long)C}, opt_f);
  auto pdg = at::empty({(long)nslabs, (long)C}, opt_f);
  auto tdb = at::empty({(long)C}, opt_f), tdg = at::empty({(long)C}, opt_f);
  auto s = bn_stream();
  DTMX_DISPATCH_16(x.scalar_type(), "bn_bwd_sums", {
    bn_bwd_stats_kernel<<<grid, 256, 0, s>>>(
        (const elem_t*)x.data_ptr(), (const elem_t*)dy.data_ptr(),
        (const elem_t*)y.data_ptr(), save_mean.data_ptr<float>(),
        save_invstd.data_ptr<float>(), pdb.data_ptr<float>(),
        pdg.data_ptr<float>(), rows, cvecs, cpb, rpb, fuse_relu ? 1 : 0);
  });
  uint32_t ncv = std::min(cvecs, 8u);
  slab_reduce2_kernel<<<(cvecs + ncv - 1) / ncv, 256, 0, s>>>(
      pdb.data_ptr<float>(), pdg.data_ptr<float>(), tdb.data_ptr<float>(),
      tdg.data_ptr<float>(), C, nslabs);
  return {tdb, tdg};
}

std::vector<at::Tensor> bn_bwd_dx_presummed(
    const at::Tensor& x, const at::Tensor& dy, const at::Tensor& y,
    const at::Tensor& save_mean, const at::Tensor& save_invstd,
    const at::Tensor& gamma, const at::Tensor& tdb, const at::Tensor& tdg,
    long count, bool fuse_relu, bool want_dres) {
  uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  uint32_t cvecs = C / 8;
  auto dx = at::empty_like(x);
  at::Tensor dres;
  if (want_dres) dres = at::empty_like(x);
  auto s = bn_stream();
  DTMX_DISPATCH_16(x.scalar_type(), "bn_bwd_dx_presummed", {
    uint32_t total8 = N * H * W * cvecs;
    FastDiv dcv;
    dcv.init(cvecs);
    uint32_t blocks = std::min<uint32_t>((total8 + 255) / 256, 2048);
    if (bn_dx_coef_on()) {
      auto opt_f = x.options().dtype(at::kFloat);
      auto ka = at::empty({(long)C}, opt_f), kb = at::empty({(long)C}, opt_f),
           kc = at::empty({(long)C}, opt_f);
      bn_dx_coef_kernel<<<(C + 255) / 256, 256, 0, s>>>(
          tdb.data_ptr<float>(), tdg.data_ptr<float>(),
          (const elem_t*)gamma.data_ptr(), save_mean.data_ptr<float>(),
          save_invstd.data_ptr<float>(), ka.data_ptr<float>(),
          kb.data_ptr<float>(), kc.data_ptr<float>(), C, 1.f / (float)count);
      launch_bn_dx_coef<elem_t>(
          (const elem_t*)x.data_ptr(), (const elem_t*)dy.data_ptr(),
          (const elem_t*)y.data_ptr(), ka.data_ptr<float>(),
          kb.data_ptr<float>(), kc.data_ptr<float>(), (elem_t*)dx.data_ptr(),
          want_dres ? (elem_t*)dres.data_ptr() : nullptr, N * H * W, cvecs,
          fuse_relu ? 1 : 0, s);
    } else {
      bn_bwd_dx_kernel<<<blocks, 256, 0, s>>>(
          (const elem_t*)x.data_ptr(), (const elem_t*)dy.data_ptr(),
          (const elem_t*)y.data_ptr(), save_mean.data_ptr<float>(),
          save_invstd.data_ptr<float>(), (const elem_t*)gamma.data_ptr(),
          tdb.data_ptr<float>(), tdg.data_ptr<float>(), (elem_t*)dx.data_ptr(),
          want_dres ? (elem_t*)dres.data_ptr() : nullptr, total8, dcv,
          1.f / (float)count, fuse_relu ? 1 : 0);
    }
  });
  if (want_dres) return {dx, dres};
  return {dx};
}

}  // namespace dtmx
