// NHWC pooling + elementwise kernels for gfx950
// (reference src/operator/nn/pool.cuh, mshadow_op elementwise kernels).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "dtmx_common.h"

namespace dtmx {

static hipStream_t pe_stream() { return at::hip::getCurrentHIPStream().stream(); }

// ------------------------------------------------------------- max pooling

// one thread = 8 channels (V8 loads/stores; scalar 2-B accesses measured
// ~10x off the bandwidth roofline on the stem maxpool)
template <typename elem_t>
__global__ void maxpool_fwd_kernel(const elem_t* __restrict__ x, elem_t* __restrict__ y,
                                   uint8_t* __restrict__ idx, uint32_t N, uint32_t C,
                                   uint32_t H, uint32_t W, uint32_t P, uint32_t Q,
                                   uint32_t K, int u, int pad, FastDiv dCv,
                                   FastDiv dQ_, FastDiv dPQ) {
  using V8 = typename E8<elem_t>::v8;
  const uint32_t cvecs = C / 8;
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  const uint32_t total = (uint32_t)(N * P * Q) * cvecs;
  const uint32_t stride = gridDim.x * blockDim.x;
  typedef __attribute__((ext_vector_type(8))) uint8_t u8x8;
  for (; i < total; i += stride) {
    uint32_t m = dCv.div(i), cv = dCv.mod(i, m);
    uint32_t np = dQ_.div(m), q = dQ_.mod(m, np);
    uint32_t n = dPQ.div(m), p = dQ_.div(dPQ.mod(m, n));
    float best[8];
    u8x8 besti;
#pragma unroll
    for (int e = 0; e < 8; ++e) { best[e] = -3.4e38f; besti[e] = 0; }
    for (uint32_t kh = 0; kh < K; ++kh) {
      int ih = (int)(p * u) - pad + (int)kh;
      if ((uint32_t)ih >= H) continue;
      for (uint32_t kw = 0; kw < K; ++kw) {
        int iw = (int)(q * u) - pad + (int)kw;
        if ((uint32_t)iw >= W) continue;
        V8 v = *(const V8*)(x + (((size_t)n * H + ih) * W + iw) * C + cv * 8);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float f = (float)v[e];
          if (f > best[e]) { best[e] = f; besti[e] = kh * K + kw; }
        }
      }
    }
    V8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) o[e] = (elem_t)best[e];
    size_t off = (size_t)m * C + cv * 8;
    *(V8*)(y + off) = o;
    *(u8x8*)(idx + off) = besti;
  }
}

template <typename elem_t>
__global__ void maxpool_bwd_kernel(const elem_t* __restrict__ dy,
                                   const uint8_t* __restrict__ idx,
                                   elem_t* __restrict__ dx, uint32_t N, uint32_t C,
                                   uint32_t H, uint32_t W, uint32_t P, uint32_t Q,
                                   uint32_t K, int u, int pad, FastDiv dCv,
                                   FastDiv dW2, FastDiv dHW) {
  using V8 = typename E8<elem_t>::v8;
  const uint32_t cvecs = C / 8;
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  const uint32_t total = (uint32_t)(N * H * W) * cvecs;
  const uint32_t stride = gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    uint32_t m = dCv.div(i), cv = dCv.mod(i, m);
    uint32_t hw_ = dHW.mod(m, dHW.div(m));
    uint32_t n = dHW.div(m);
    uint32_t h = dW2.div(hw_), w = dW2.mod(hw_, h);
    // windows (p,q) that contain (h,w): p*u - pad <= h < p*u - pad + K
    int plo = ((int)h + pad - (int)K + u) / u;  // ceil((h+pad-K+1)/u)
    if (plo < 0) plo = 0;
    int phi = ((int)h + pad) / u;
    if (phi >= (int)P) phi = P - 1;
    int qlo = ((int)w + pad - (int)K + u) / u;
    if (qlo < 0) qlo = 0;
    int qhi = ((int)w + pad) / u;
    if (qhi >= (int)Q) qhi = Q - 1;
    typedef __attribute__((ext_vector_type(8))) uint8_t u8x8;
    float acc8[8] = {};
    for (int p = plo; p <= phi; ++p) {
      uint32_t kh = (uint32_t)((int)h + pad - p * u);
      if (kh >= K) continue;
      for (int q = qlo; q <= qhi; ++q) {
        uint32_t kw = (uint32_t)((int)w + pad - q * u);
        if (kw >= K) continue;
        size_t o = (((size_t)n * P + p) * Q + q) * C + cv * 8;
        u8x8 iv = *(const u8x8*)(idx + o);
        V8 gv = *(const V8*)(dy + o);
        uint8_t want = (uint8_t)(kh * K + kw);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          if (iv[e] == want) acc8[e] += (float)gv[e];
      }
    }
    V8 o8;
#pragma unroll
    for (int e = 0; e < 8; ++e) o8[e] = (elem_t)acc8[e];
    *(V8*)(dx + (size_t)m * C + cv * 8) = o8;
  }
}

// scalar fallbacks for C % 8 != 0 (zoo models like lenet's 20 channels)
template <typename elem_t>
__global__ void maxpool_fwd_scalar_kernel(const elem_t* __restrict__ x,
                                          elem_t* __restrict__ y,
                                          uint8_t* __restrict__ idx, uint32_t N,
                                          uint32_t C, uint32_t H, uint32_t W,
                                          uint32_t P, uint32_t Q, uint32_t K,
                                          int u, int pad) {
  using V8 = typename E8<elem_t>::v8;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t total = (size_t)N * P * Q * C;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    uint32_t c = i % C;
    size_t m = i / C;
    uint32_t q = m % Q;
    uint32_t p = (m / Q) % P;
    uint32_t n = m / ((size_t)P * Q);
    float best = -3.4e38f;
    uint8_t besti = 0;
    for (uint32_t kh = 0; kh < K; ++kh) {
      int ih = (int)(p * u) - pad + (int)kh;
      if ((uint32_t)ih >= H) continue;
      for (uint32_t kw = 0; kw < K; ++kw) {
        int iw = (int)(q * u) - pad + (int)kw;
        if ((uint32_t)iw >= W) continue;
        float v = (float)x[(((size_t)n * H + ih) * W + iw) * C + c];
        if (v > best) { best = v; besti = kh * K + kw; }
      }
    }
    y[i] = (elem_t)best;
    idx[i] = besti;
  }
}

template <typename elem_t>
__global__ void maxpool_bwd_scalar_kernel(const elem_t* __restrict__ dy,
                                          const uint8_t* __restrict__ idx,
                                          elem_t* __restrict__ dx, uint32_t N,
                                          uint32_t C, uint32_t H, uint32_t W,
                                          uint32_t P, uint32_t Q, uint32_t K,
                                          int u, int pad) {
  using V8 = typename E8<elem_t>::v8;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t total = (size_t)N * H * W * C;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    uint32_t c = i % C;
    size_t m = i / C;
    uint32_t w = m % W;
    uint32_t h = (m / W) % H;
    uint32_t n = m / ((size_t)H * W);
    int plo = ((int)h + pad - (int)K + u) / u;
    if (plo < 0) plo = 0;
    int phi = ((int)h + pad) / u;
    if (phi >= (int)P) phi = P - 1;
    int qlo = ((int)w + pad - (int)K + u) / u;
    if (qlo < 0) qlo = 0;
    int qhi = ((int)w + pad) / u;
    if (qhi >= (int)Q) qhi = Q - 1;
    float acc = 0.f;
    for (int p = plo; p <= phi; ++p) {
      uint32_t kh = (uint32_t)((int)h + pad - p * u);
      if (kh >= K) continue;
      for (int q = qlo; q <= qhi; ++q) {
        uint32_t kw = (uint32_t)((int)w + pad - q * u);
        if (kw >= K) continue;
        size_t o = (((size_t)n * P + p) * Q + q) * C + c;
        if (idx[o] == kh * K + kw) acc += (float)dy[o];
      }
    }
    dx[i] = (elem_t)acc;
  }
}

// ------------------------------------------------------- global average pool

template <typename elem_t>
__global__ void gap_fwd_kernel(const elem_t* __restrict__ x, elem_t* __restrict__ y,
                               uint32_t N, uint32_t C, uint32_t HW) {
  using V8 = typename E8<elem_t>::v8;
  uint32_t c = blockIdx.x * blockDim.x + threadIdx.x;
  uint32_t n = blockIdx.y;
  if (c >= C) return;
  const elem_t* base = x + (size_t)n * HW * C + c;
  float s = 0.f;
  for (uint32_t i = 0; i < HW; ++i) s += (float)base[(size_t)i * C];
  y[(size_t)n * C + c] = (elem_t)(s / HW);
}

template <typename elem_t>
__global__ void gap_bwd_kernel(const elem_t* __restrict__ dy, elem_t* __restrict__ dx,
                               uint32_t N, uint32_t C, uint32_t HW) {
  using V8 = typename E8<elem_t>::v8;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t total = (size_t)N * HW * C;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  float inv = 1.f / HW;
  for (; i < total; i += stride) {
    uint32_t c = i % C;
    uint32_t n = i / ((size_t)HW * C);
    dx[i] = (elem_t)((float)dy[(size_t)n * C + c] * inv);
  }
}

// -------------------------------------------------------------- elementwise

template <typename elem_t>
__global__ void relu_fwd_kernel(const elem_t* __restrict__ x, elem_t* __restrict__ y,
                                size_t total8) {
  using V8 = typename E8<elem_t>::v8;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total8; i += stride) {
    V8 v = *(const V8*)(x + i * 8);
    V8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) o[e] = (elem_t)fmaxf((float)v[e], 0.f);
    *(V8*)(y + i * 8) = o;
  }
}

template <typename elem_t>
__global__ void relu_bwd_kernel(const elem_t* __restrict__ dy,
                                const elem_t* __restrict__ y,
                                elem_t* __restrict__ dx, size_t total8) {
  using V8 = typename E8<elem_t>::v8;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total8; i += stride) {
    V8 g = *(const V8*)(dy + i * 8);
    V8 v = *(const V8*)(y + i * 8);
    V8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      o[e] = (float)v[e] > 0.f ? g[e] : (elem_t)0.f;
    *(V8*)(dx + i * 8) = o;
  }
}

template <typename elem_t>
__global__ void add_relu_kernel(const elem_t* __restrict__ a,
                                const elem_t* __restrict__ b,
                                elem_t* __restrict__ y, size_t total8) {
  using V8 = typename E8<elem_t>::v8;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total8; i += stride) {
    V8 va = *(const V8*)(a + i * 8);
    V8 vb = *(const V8*)(b + i * 8);
    V8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      o[e] = (elem_t)fmaxf((float)va[e] + (float)vb[e], 0.f);
    *(V8*)(y + i * 8) = o;
  }
}

// ============================================================== host side ==

static uint32_t ew_blocks(size_t work) {
  return std::min<size_t>((work + 255) / 256, 2048);
}

std::vector<at::Tensor> maxpool_fwd(const at::Tensor& x, long kernel, long stride,
                                    long pad) {
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "maxpool: x must be NHWC");
  uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  uint32_t P = (H + 2 * pad - kernel) / stride + 1;
  uint32_t Q = (W + 2 * pad - kernel) / stride + 1;
  auto y = at::empty({(long)N, (long)C, (long)P, (long)Q}, x.options(),
                     at::MemoryFormat::ChannelsLast);
  auto idx = at::empty({(long)N, (long)C, (long)P, (long)Q},
                       x.options().dtype(at::kByte), at::MemoryFormat::ChannelsLast);
  DTMX_DISPATCH_16(x.scalar_type(), "maxpool", {
    if (C % 8 == 0) {
      FastDiv dCv, dQ_, dPQ;
      dCv.init(C / 8); dQ_.init(Q); dPQ.init(P * Q);
      size_t total = (size_t)N * P * Q * (C / 8);
      maxpool_fwd_kernel<<<ew_blocks(total), 256, 0, pe_stream()>>>(
          (const elem_t*)x.data_ptr(), (elem_t*)y.data_ptr(),
          (uint8_t*)idx.data_ptr(), N, C, H, W, P, Q, kernel, stride, pad,
          dCv, dQ_, dPQ);
    } else {
      size_t total = (size_t)N * P * Q * C;
      maxpool_fwd_scalar_kernel<<<ew_blocks(total), 256, 0, pe_stream()>>>(
          (const elem_t*)x.data_ptr(), (elem_t*)y.data_ptr(),
          (uint8_t*)idx.data_ptr(), N, C, H, W, P, Q, kernel, stride, pad);
    }
  });
  return {y, idx};
}

at::Tensor maxpool_bwd(const at::Tensor& dy, const at::Tensor& idx, long H,
                            long W, long kernel, long stride, long pad) {
  uint32_t N = dy.size(0), C = dy.size(1), P = dy.size(2), Q = dy.size(3);
  auto dyc = dy.contiguous(at::MemoryFormat::ChannelsLast);
  auto dx = at::empty({(long)N, (long)C, (long)H, (long)W}, dy.options(),
                      at::MemoryFormat::ChannelsLast);
  DTMX_DISPATCH_16(dy.scalar_type(), "maxpool_bwd", {
    if (C % 8 == 0) {
      FastDiv dCv, dW2, dHW;
      dCv.init(C / 8); dW2.init(W); dHW.init(H * W);
      size_t total = (size_t)N * H * W * (C / 8);
      maxpool_bwd_kernel<<<ew_blocks(total), 256, 0, pe_stream()>>>(
          (const elem_t*)dyc.data_ptr(), (const uint8_t*)idx.data_ptr(),
          (elem_t*)dx.data_ptr(), N, C, H, W, P, Q, kernel, stride, pad,
          dCv, dW2, dHW);
    } else {
      size_t total = (size_t)N * H * W * C;
      maxpool_bwd_scalar_kernel<<<ew_blocks(total), 256, 0, pe_stream()>>>(
          (const elem_t*)dyc.data_ptr(), (const uint8_t*)idx.data_ptr(),
          (elem_t*)dx.data_ptr(), N, C, H, W, P, Q, kernel, stride, pad);
    }
  });
  return dx;
}

at::Tensor global_avgpool_fwd(const at::Tensor& x) {
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "gap: x must be NHWC");
  uint32_t N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto y = at::empty({(long)N, (long)C}, x.options());
  dim3 grid((C + 255) / 256, N);
  DTMX_DISPATCH_16(x.scalar_type(), "gap", {
    gap_fwd_kernel<<<grid, 256, 0, pe_stream()>>>(
        (const elem_t*)x.data_ptr(), (elem_t*)y.data_ptr(), N, C, HW);
  });
  return y;
}

at::Tensor global_avgpool_bwd(const at::Tensor& dy, long H, long W) {
  uint32_t N = dy.size(0), C = dy.size(1), HW = H * W;
  auto dx = at::empty({(long)N, (long)C, H, W}, dy.options(),
                      at::MemoryFormat::ChannelsLast);
  size_t total = (size_t)N * HW * C;
  DTMX_DISPATCH_16(dy.scalar_type(), "gap_bwd", {
    gap_bwd_kernel<<<ew_blocks(total), 256, 0, pe_stream()>>>(
        (const elem_t*)dy.data_ptr(), (elem_t*)dx.data_ptr(), N, C, HW);
  });
  return dx;
}

#define EW_CHECK(t)                                                        \
  TORCH_CHECK((t).scalar_type() == at::kBFloat16 ||                        \
                  (t).scalar_type() == at::kHalf,                          \
              "16-bit float only");                                        \
  TORCH_CHECK((t).numel() % 8 == 0, "numel must be a multiple of 8")

at::Tensor relu_fwd(const at::Tensor& x) {
  EW_CHECK(x);
  auto y = at::empty_like(x);
  size_t t8 = x.numel() / 8;
  DTMX_DISPATCH_16(x.scalar_type(), "relu", {
    relu_fwd_kernel<<<ew_blocks(t8), 256, 0, pe_stream()>>>(
        (const elem_t*)x.data_ptr(), (elem_t*)y.data_ptr(), t8);
  });
  return y;
}

at::Tensor relu_bwd(const at::Tensor& dy, const at::Tensor& y) {
  EW_CHECK(dy);
  auto dx = at::empty_like(dy);
  size_t t8 = dy.numel() / 8;
  DTMX_DISPATCH_16(dy.scalar_type(), "relu_bwd", {
    relu_bwd_kernel<<<ew_blocks(t8), 256, 0, pe_stream()>>>(
        (const elem_t*)dy.data_ptr(), (const elem_t*)y.data_ptr(),
        (elem_t*)dx.data_ptr(), t8);
  });
  return dx;
}

at::Tensor add_relu_fwd(const at::Tensor& a, const at::Tensor& b) {
  EW_CHECK(a);
  auto y = at::empty_like(a);
  size_t t8 = a.numel() / 8;
  DTMX_DISPATCH_16(a.scalar_type(), "add_relu", {
    add_relu_kernel<<<ew_blocks(t8), 256, 0, pe_stream()>>>(
        (const elem_t*)a.data_ptr(), (const elem_t*)b.data_ptr(),
        (elem_t*)y.data_ptr(), t8);
  });
  return y;
}

// ---------------------------------------------- LRN (cross-channel, n=5)
// Reference src/operator/nn/lrn.cc (mx.sym.LRN): y = x * s^-beta with
// s = knorm + alpha/n * sum_{|c'-c|<=2} x_{c'}^2. NHWC channels are
// contiguous, so a thread owns one (row, 8-channel vector) and reads +-1
// vector of halo; out-of-range halo loads return 0, which IS the clamped
// window (summing zero == not summing). The torch lowering
// (pad+avg_pool3d) measured 3.3x-of-step on alexnet; this is one
// streaming pass each way.
template <typename elem_t>
__global__ void lrn_fwd_kernel(const elem_t* __restrict__ x,
                               elem_t* __restrict__ y, uint32_t rows,
                               uint32_t cvecs, float an, float beta,
                               float knorm) {
  using V8 = typename E8<elem_t>::v8;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t total = (size_t)rows * cvecs;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    uint32_t cv = (uint32_t)(i % cvecs);
    V8 xc = *(const V8*)(x + i * 8);
    V8 xl = {}, xr = {};
    if (cv > 0) xl = *(const V8*)(x + (i - 1) * 8);
    if (cv + 1 < cvecs) xr = *(const V8*)(x + (i + 1) * 8);
    float sq[12];  // squared channels cv*8-2 .. cv*8+9
#pragma unroll
    for (int e = 0; e < 2; ++e) {
      float v = (float)xl[6 + e];
      sq[e] = v * v;
    }
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float v = (float)xc[e];
      sq[2 + e] = v * v;
    }
#pragma unroll
    for (int e = 0; e < 2; ++e) {
      float v = (float)xr[e];
      sq[10 + e] = v * v;
    }
    V8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float win = sq[e] + sq[e + 1] + sq[e + 2] + sq[e + 3] + sq[e + 4];
      float s = knorm + an * win;
      o[e] = (elem_t)((float)xc[e] * __powf(s, -beta));
    }
    *(V8*)(y + i * 8) = o;
  }
}

// dx_c = dy_c * s_c^-beta - (2*alpha*beta/n) * x_c *
//        sum_{c' in win(c)} dy_{c'} * x_{c'} * s_{c'}^(-beta-1)
template <typename elem_t>
__global__ void lrn_bwd_kernel(const elem_t* __restrict__ x,
                               const elem_t* __restrict__ dy,
                               elem_t* __restrict__ dx, uint32_t rows,
                               uint32_t cvecs, float an, float beta,
                               float knorm) {
  using V8 = typename E8<elem_t>::v8;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t total = (size_t)rows * cvecs;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    uint32_t cv = (uint32_t)(i % cvecs);
    // x halo +-1 vector covers channels cv*8-8 .. cv*8+15: the s-windows
    // of the +-2 neighbor channels need x out to +-4; dy halo +-2 channels
    float xs[24];
    float gs[12];
#pragma unroll
    for (int e = 0; e < 24; ++e) xs[e] = 0.f;
#pragma unroll
    for (int e = 0; e < 12; ++e) gs[e] = 0.f;
    V8 xc = *(const V8*)(x + i * 8);
    V8 gc = *(const V8*)(dy + i * 8);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      xs[8 + e] = (float)xc[e];
      gs[2 + e] = (float)gc[e];
    }
    if (cv > 0) {
      V8 v = *(const V8*)(x + (i - 1) * 8);
      V8 g = *(const V8*)(dy + (i - 1) * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e) xs[e] = (float)v[e];
#pragma unroll
      for (int e = 0; e < 2; ++e) gs[e] = (float)g[6 + e];
    }
    if (cv + 1 < cvecs) {
      V8 v = *(const V8*)(x + (i + 1) * 8);
      V8 g = *(const V8*)(dy + (i + 1) * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e) xs[16 + e] = (float)v[e];
#pragma unroll
      for (int e = 0; e < 2; ++e) gs[10 + e] = (float)g[e];
    }
    // t[j] = dy_j * x_j * s_j^(-beta-1) for the 12 channels cv*8-2..cv*8+9
    // (channel cv*8+d sits at xs[8+d] / gs[2+d]; j = d+2)
    float t[12];
#pragma unroll
    for (int j = 0; j < 12; ++j) {
      float win = 0.f;
#pragma unroll
      for (int w = 0; w < 5; ++w) {
        float v = xs[j + 4 + w];  // channel (j-2) + (w-2) -> xs[8+j-2+w-2]
        win += v * v;
      }
      float s = knorm + an * win;
      t[j] = gs[j] * xs[j + 6] * __powf(s, -beta - 1.f);
    }
    V8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float win = 0.f;
#pragma unroll
      for (int w = 0; w < 5; ++w) {
        float v = xs[e + 6 + w];
        win += v * v;
      }
      float s = knorm + an * win;
      float tsum = t[e] + t[e + 1] + t[e + 2] + t[e + 3] + t[e + 4];
      o[e] = (elem_t)(gs[2 + e] * __powf(s, -beta) -
                      2.f * an * beta * xs[8 + e] * tsum);
    }
    *(V8*)(dx + i * 8) = o;
  }
}

at::Tensor lrn_fwd(const at::Tensor& x, double alpha, double beta,
                   double knorm) {
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "lrn: NHWC");
  uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(C % 8 == 0, "lrn: C % 8 != 0");
  auto y = at::empty_like(x);
  size_t total = (size_t)N * H * W * (C / 8);
  DTMX_DISPATCH_16(x.scalar_type(), "lrn_fwd", {
    lrn_fwd_kernel<<<ew_blocks(total), 256, 0, pe_stream()>>>(
        (const elem_t*)x.data_ptr(), (elem_t*)y.data_ptr(), N * H * W, C / 8,
        (float)(alpha / 5.0), (float)beta, (float)knorm);
  });
  return y;
}

at::Tensor lrn_bwd(const at::Tensor& x, const at::Tensor& dy, double alpha,
                   double beta, double knorm) {
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast), "lrn: NHWC");
  uint32_t N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  auto dyc = dy.contiguous(at::MemoryFormat::ChannelsLast);
  auto dx = at::empty_like(x);
  size_t total = (size_t)N * H * W * (C / 8);
  DTMX_DISPATCH_16(x.scalar_type(), "lrn_bwd", {
    lrn_bwd_kernel<<<ew_blocks(total), 256, 0, pe_stream()>>>(
        (const elem_t*)x.data_ptr(), (const elem_t*)dyc.data_ptr(),
        (elem_t*)dx.data_ptr(), N * H * W, C / 8, (float)(alpha / 5.0),
        (float)beta, (float)knorm);
  });
  return dx;
}

}  // namespace dtmx
