// Embedding/take gather + scatter-add for gfx950 (reference
// src/operator/tensor/indexing_op.cu Take/AddTakeGrad — redesigned for
// NHWC-free 2-D tables, V8-vectorized rows, fp32 scatter accumulation).
//
// take_fwd:   out[i][:] = table[idx[i]][:]         (Embedding forward)
// take_bwd:   dtab[r]  += sum_{i: idx[i]==r} dy[i] (fp32 atomics; `idx` may
//             be pre-compacted via unique+inverse for a row-sparse [U][D]
//             accumulator — the kernel is the same either way)
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "dtmx_common.h"

namespace dtmx {

static hipStream_t idx_stream() { return at::hip::getCurrentHIPStream().stream(); }

template <typename elem_t>
__global__ void take_fwd_kernel(const elem_t* __restrict__ table,
                                const int64_t* __restrict__ idx,
                                elem_t* __restrict__ out, uint32_t N,
                                uint32_t D8, uint32_t V) {
  using V8 = typename E8<elem_t>::v8;
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t total = (size_t)N * D8;
  for (; i < total; i += stride) {
    uint32_t row = i / D8, c = i % D8;
    int64_t r = idx[row];
    // reference take default mode 'clip': clamp OOB indices
    r = r < 0 ? 0 : (r >= (int64_t)V ? (int64_t)V - 1 : r);
    *(V8*)(out + ((size_t)row * D8 + c) * 8) =
        *(const V8*)(table + ((size_t)r * D8 + c) * 8);
  }
}

template <typename elem_t>
__global__ void take_bwd_kernel(const elem_t* __restrict__ dy,
                                const int64_t* __restrict__ idx,
                                float* __restrict__ dtab, uint32_t N,
                                uint32_t D, uint32_t V) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t total = (size_t)N * D;
  for (; i < total; i += stride) {
    uint32_t row = i / D, c = i % D;
    int64_t r = idx[row];
    if (r >= 0 && r < (int64_t)V)
      atomicAdd(&dtab[(size_t)r * D + c], (float)dy[i]);
  }
}

at::Tensor take_fwd(const at::Tensor& table, const at::Tensor& idx) {
  DTMX_DISPATCH_16(table.scalar_type(), "take_fwd", {
    TORCH_CHECK(table.is_cuda() && table.dim() == 2 && table.is_contiguous(),
                "take_fwd: table must be contiguous 2-D CUDA");
    TORCH_CHECK(idx.scalar_type() == at::kLong, "take_fwd: idx must be int64");
    uint32_t V = table.size(0), D = table.size(1);
    TORCH_CHECK(D % 8 == 0, "take_fwd: embedding dim must be a multiple of 8");
    auto idxc = idx.contiguous();
    uint32_t N = idxc.numel();
    auto out = at::empty({(long)N, (long)D}, table.options());
    size_t total = (size_t)N * (D / 8);
    uint32_t blocks = std::min<size_t>((total + 255) / 256, 8192);
    take_fwd_kernel<<<blocks, 256, 0, idx_stream()>>>(
        (const elem_t*)table.data_ptr(), idxc.data_ptr<int64_t>(),
        (elem_t*)out.data_ptr(), N, D / 8, V);
    return out;
  });
  return at::Tensor();
}

// dy:[N][D], idx:[N] -> fp32 [V][D] accumulator (V = target row count; pass
// the compact unique-row count + inverse indices for a row-sparse grad)
at::Tensor take_bwd(const at::Tensor& dy, const at::Tensor& idx, long V) {
  DTMX_DISPATCH_16(dy.scalar_type(), "take_bwd", {
    TORCH_CHECK(dy.is_cuda() && dy.dim() == 2, "take_bwd: dy must be 2-D CUDA");
    auto dyc = dy.contiguous();
    auto idxc = idx.contiguous();
    uint32_t N = dyc.size(0), D = dyc.size(1);
    auto dtab = at::zeros({V, (long)D}, dy.options().dtype(at::kFloat));
    size_t total = (size_t)N * D;
    uint32_t blocks = std::min<size_t>((total + 255) / 256, 8192);
    take_bwd_kernel<<<blocks, 256, 0, idx_stream()>>>(
        (const elem_t*)dyc.data_ptr(), idxc.data_ptr<int64_t>(),
        dtab.data_ptr<float>(), N, D, V);
    return dtab;
  });
  return at::Tensor();
}

}  // namespace dtmx
