// Fused softmax + cross-entropy (reference src/operator/nn/softmax-inl.h:
// 166-260 + softmax_output-inl.h fused loss-grad op) and the fused
// multi-precision SGD/momentum update (reference optimizer_op-inl.h:86,305,430)
// for gfx950.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "dtmx_common.h"

namespace dtmx {

static hipStream_t so_stream() { return at::hip::getCurrentHIPStream().stream(); }

// one block (256 threads) per row: max -> exp-sum -> probs + per-row loss
template <typename elem_t>
__global__ void softmax_ce_fwd_kernel(const elem_t* __restrict__ logits,
                                      const int* __restrict__ label,
                                      elem_t* __restrict__ probs,
                                      float* __restrict__ loss, uint32_t V) {
  const uint32_t row = blockIdx.x;
  const uint32_t t = threadIdx.x;
  const elem_t* in = logits + (size_t)row * V;
  elem_t* out = probs + (size_t)row * V;
  __shared__ float red[8];

  float m = -3.4e38f;
  for (uint32_t i = t; i < V; i += 256) m = fmaxf(m, (float)in[i]);
#pragma unroll
  for (int o = 32; o > 0; o >>= 1) m = fmaxf(m, __shfl_down(m, o));
  if ((t & 63) == 0) red[t >> 6] = m;
  __syncthreads();
  m = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));

  float s = 0.f;
  for (uint32_t i = t; i < V; i += 256) s += __expf((float)in[i] - m);
#pragma unroll
  for (int o = 32; o > 0; o >>= 1) s += __shfl_down(s, o);
  if ((t & 63) == 0) red[4 + (t >> 6)] = s;
  __syncthreads();
  s = red[4] + red[5] + red[6] + red[7];
  const float inv = 1.f / s, logs = __logf(s);

  for (uint32_t i = t; i < V; i += 256)
    out[i] = (elem_t)(__expf((float)in[i] - m) * inv);
  if (t == 0) {
    int y = label[row];
    float lp = ((float)in[y] - m) - logs;  // log softmax at the label
    atomicAdd(loss, -lp);
  }
}

// d_logits = (p - onehot) * dloss   (sum-CE; reference SoftmaxOutput grad)
template <typename elem_t>
__global__ void softmax_ce_bwd_kernel(const elem_t* __restrict__ probs,
                                      const int* __restrict__ label,
                                      const float* __restrict__ dloss,
                                      elem_t* __restrict__ dlogits, uint32_t V,
                                      size_t total) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const float g = *dloss;
  for (; i < total; i += stride) {
    uint32_t row = i / V, col = i % V;
    float p = (float)probs[i];
    if ((int)col == label[row]) p -= 1.f;
    dlogits[i] = (elem_t)(p * g);
  }
}

// ---- fused SGD/momentum, multi-precision (bf16 weights + fp32 master) ----
//   g32  = clip(grad * rescale) + wd * master
//   mom  = momentum * mom - lr * g32
//   master += mom;  w = bf16(master)
template <typename elem_t>
__global__ void sgd_mom_mp_kernel(elem_t* __restrict__ w, const elem_t* __restrict__ g,
                                  float* __restrict__ master, float* __restrict__ mom,
                                  size_t total, float lr, float momentum, float wd,
                                  float rescale, float clip) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    float gv = (float)g[i] * rescale;
    if (clip > 0.f) gv = fminf(fmaxf(gv, -clip), clip);
    gv += wd * master[i];
    float m = momentum * mom[i] - lr * gv;
    mom[i] = m;
    float nw = master[i] + m;
    master[i] = nw;
    w[i] = (elem_t)nw;
  }
}

// device-hyperparameter variant (hipGraph-capturable: lr/rescale live in a
// device buffer updated between replays, not baked into kernel args).
// hyper = {lr, momentum, wd, rescale, clip}
template <typename elem_t>
__global__ void sgd_mom_mp_dev_kernel(elem_t* __restrict__ w,
                                      const elem_t* __restrict__ g,
                                      float* __restrict__ master,
                                      float* __restrict__ mom, size_t total,
                                      const float* __restrict__ hyper) {
  const float lr = hyper[0], momentum = hyper[1], wd = hyper[2],
              rescale = hyper[3], clip = hyper[4];
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    float gv = (float)g[i] * rescale;
    if (clip > 0.f) gv = fminf(fmaxf(gv, -clip), clip);
    gv += wd * master[i];
    float m = momentum * mom[i] - lr * gv;
    mom[i] = m;
    float nw = master[i] + m;
    master[i] = nw;
    w[i] = (elem_t)nw;
  }
}

__global__ void sgd_mom_f32_kernel(float* __restrict__ w, const float* __restrict__ g,
                                   float* __restrict__ mom, size_t total, float lr,
                                   float momentum, float wd, float rescale,
                                   float clip) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    float gv = g[i] * rescale;
    if (clip > 0.f) gv = fminf(fmaxf(gv, -clip), clip);
    gv += wd * w[i];
    float m = momentum * mom[i] - lr * gv;
    mom[i] = m;
    w[i] += m;
  }
}

// ============================================================== host side ==

std::vector<at::Tensor> softmax_ce_fwd(const at::Tensor& logits,
                                       const at::Tensor& label) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2, "softmax_ce: 2D logits");
  auto lc = logits.contiguous();
  auto yc = label.to(at::kInt).contiguous();
  uint32_t B = lc.size(0), V = lc.size(1);
  auto probs = at::empty_like(lc);
  auto loss = at::zeros({}, lc.options().dtype(at::kFloat));
  DTMX_DISPATCH_16(lc.scalar_type(), "softmax_ce", {
    softmax_ce_fwd_kernel<<<B, 256, 0, so_stream()>>>(
        (const elem_t*)lc.data_ptr(), yc.data_ptr<int>(),
        (elem_t*)probs.data_ptr(), loss.data_ptr<float>(), V);
  });
  return {loss, probs};
}

at::Tensor softmax_ce_bwd(const at::Tensor& probs, const at::Tensor& label,
                          const at::Tensor& dloss) {
  auto yc = label.to(at::kInt).contiguous();
  uint32_t B = probs.size(0), V = probs.size(1);
  auto dl = at::empty_like(probs);
  auto dlf = dloss.to(at::kFloat).contiguous();
  size_t total = (size_t)B * V;
  uint32_t blocks = std::min<size_t>((total + 255) / 256, 2048);
  DTMX_DISPATCH_16(probs.scalar_type(), "softmax_ce_bwd", {
    softmax_ce_bwd_kernel<<<blocks, 256, 0, so_stream()>>>(
        (const elem_t*)probs.data_ptr(), yc.data_ptr<int>(),
        dlf.data_ptr<float>(), (elem_t*)dl.data_ptr(), V, total);
  });
  return dl;
}

void sgd_mom_mp(at::Tensor w, const at::Tensor& g, at::Tensor master,
                at::Tensor mom, double lr, double momentum, double wd,
                double rescale, double clip) {
  size_t total = w.numel();
  uint32_t blocks = std::min<size_t>((total + 255) / 256, 4096);
  DTMX_DISPATCH_16(w.scalar_type(), "sgd_mom_mp", {
    sgd_mom_mp_kernel<<<blocks, 256, 0, so_stream()>>>(
        (elem_t*)w.data_ptr(), (const elem_t*)g.data_ptr(),
        master.data_ptr<float>(), mom.data_ptr<float>(), total, lr, momentum, wd,
        rescale, clip);
  });
}

void sgd_mom_mp_dev(at::Tensor w, const at::Tensor& g, at::Tensor master,
                    at::Tensor mom, const at::Tensor& hyper) {
  size_t total = w.numel();
  uint32_t blocks = std::min<size_t>((total + 255) / 256, 4096);
  DTMX_DISPATCH_16(w.scalar_type(), "sgd_mom_mp_dev", {
    sgd_mom_mp_dev_kernel<<<blocks, 256, 0, so_stream()>>>(
        (elem_t*)w.data_ptr(), (const elem_t*)g.data_ptr(),
        master.data_ptr<float>(), mom.data_ptr<float>(), total,
        hyper.data_ptr<float>());
  });
}

void sgd_mom_f32(at::Tensor w, const at::Tensor& g, at::Tensor mom, double lr,
                 double momentum, double wd, double rescale, double clip) {
  size_t total = w.numel();
  uint32_t blocks = std::min<size_t>((total + 255) / 256, 4096);
  sgd_mom_f32_kernel<<<blocks, 256, 0, so_stream()>>>(
      w.data_ptr<float>(), g.data_ptr<float>(), mom.data_ptr<float>(), total,
      lr, momentum, wd, rescale, clip);
}

}  // namespace dtmx
