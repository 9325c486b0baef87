// 2-bit gradient compression with error-feedback residual for gfx950
// (reference src/kvstore/gradient_compression-inl.h:40-135,
// gradient_compression.cu:29-39). Wire format: 16 two-bit codes per uint32,
// 01 = +threshold, 10 = -threshold, 00 = zero; residual accumulates the
// quantization error on the worker.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "dtmx_common.h"

namespace dtmx {

static hipStream_t cc_stream() { return at::hip::getCurrentHIPStream().stream(); }

template <typename elem_t>
__global__ void quantize_2bit_kernel(const elem_t* __restrict__ grad,
                                     float* __restrict__ residual,
                                     uint32_t* __restrict__ out, size_t words,
                                     size_t n, float threshold) {
  size_t w = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; w < words; w += stride) {
    uint32_t packed = 0;
    size_t base = w * 16;
#pragma unroll 4
    for (int e = 0; e < 16; ++e) {
      size_t i = base + e;
      if (i >= n) break;
      float g = (float)grad[i] + residual[i];
      uint32_t code = 0;
      float q = 0.f;
      if (g >= threshold) {
        code = 1;
        q = threshold;
      } else if (g <= -threshold) {
        code = 2;
        q = -threshold;
      }
      residual[i] = g - q;
      packed |= code << (2 * e);
    }
    out[w] = packed;
  }
}

template <typename elem_t>
__global__ void dequantize_2bit_kernel(const uint32_t* __restrict__ in,
                                       elem_t* __restrict__ out, size_t words,
                                       size_t n, float threshold) {
  size_t w = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; w < words; w += stride) {
    uint32_t packed = in[w];
    size_t base = w * 16;
#pragma unroll 4
    for (int e = 0; e < 16; ++e) {
      size_t i = base + e;
      if (i >= n) break;
      uint32_t code = (packed >> (2 * e)) & 0x3;
      out[i] = (elem_t)(code == 1 ? threshold : code == 2 ? -threshold : 0.f);
    }
  }
}

at::Tensor quantize_2bit(const at::Tensor& grad, at::Tensor residual,
                         double threshold) {
  TORCH_CHECK(grad.is_cuda(), "quantize_2bit: CUDA tensor required");
  size_t n = grad.numel();
  size_t words = (n + 15) / 16;
  auto out = at::empty({(long)words}, grad.options().dtype(at::kInt));
  uint32_t blocks = std::min<size_t>((words + 255) / 256, 2048);
  DTMX_DISPATCH_16(grad.scalar_type(), "quantize_2bit", {
    quantize_2bit_kernel<<<blocks, 256, 0, cc_stream()>>>(
        (const elem_t*)grad.data_ptr(), residual.data_ptr<float>(),
        (uint32_t*)out.data_ptr(), words, n, threshold);
  });
  return out;
}

at::Tensor dequantize_2bit(const at::Tensor& packed, long numel, double threshold) {
  size_t words = packed.numel();
  auto out = at::empty({numel}, packed.options().dtype(at::kBFloat16));
  uint32_t blocks = std::min<size_t>((words + 255) / 256, 2048);
  DTMX_DISPATCH_16(out.scalar_type(), "dequantize_2bit", {
    dequantize_2bit_kernel<<<blocks, 256, 0, cc_stream()>>>(
        (const uint32_t*)packed.data_ptr(), (elem_t*)out.data_ptr(), words,
        numel, threshold);
  });
  return out;
}

}  // namespace dtmx
