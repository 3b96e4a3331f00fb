// K8: fused cross-entropy over the vocab — streamed online log-softmax
// (single pass over logits: running max + rescaled sum), no fp32
// logits-sized temporary. Replaces reference nn.CrossEntropyLoss
// (loss_functions.py:33-52) which materializes fp32 softmax over
// [B*T, 50304].
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr int BLOCK = 256;

// one workgroup per row; online max/sum in one pass over V.
__global__ void ce_fwd_kernel(const unsigned short* __restrict__ logits,
                              const long* __restrict__ targets,
                              float* __restrict__ losses,
                              float* __restrict__ lse_out,
                              int V, long ignore_index) {
  __shared__ float scratch[BLOCK / WAVE_SIZE];
  const long row = blockIdx.x;
  const long tgt = targets[row];
  const unsigned short* lr = logits + row * (long)V;

  float m = -INFINITY, s = 0.f;
  const int vecV = V / 8;
  const shortx8* lv = reinterpret_cast<const shortx8*>(lr);
  for (int i = threadIdx.x; i < vecV; i += BLOCK) {
    shortx8 v = lv[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32((unsigned short)v[j]);
      if (f > m) { s *= __expf(m - f); m = f; }
      s += __expf(f - m);
    }
  }
  for (int i = vecV * 8 + threadIdx.x; i < V; i += BLOCK) {
    float f = bf16_to_f32(lr[i]);
    if (f > m) { s *= __expf(m - f); m = f; }
    s += __expf(f - m);
  }
  // combine per-thread (m, s) pairs: block max then rescaled sums
  float bm = block_reduce_max(m, scratch);
  s *= __expf(m - bm);
  float bs = block_reduce_sum(s, scratch);
  const float lse = bm + __logf(bs);
  if (threadIdx.x == 0) {
    lse_out[row] = lse;
    losses[row] = (tgt == ignore_index) ? 0.f
                  : lse - bf16_to_f32(lr[tgt]);
  }
}

// dlogits = scale * (softmax - onehot); rows with ignore_index -> 0.
__global__ void ce_bwd_kernel(const unsigned short* __restrict__ logits,
                              const long* __restrict__ targets,
                              const float* __restrict__ lse,
                              unsigned short* __restrict__ dlogits,
                              const float* __restrict__ scale_ptr, int V,
                              long ignore_index) {
  const float scale = *scale_ptr;  // device-resident: hipGraph-capture-safe
  const long row = blockIdx.x;
  const long tgt = targets[row];
  const unsigned short* lr = logits + row * (long)V;
  unsigned short* dr = dlogits + row * (long)V;
  const float l = lse[row];
  const bool ignored = (tgt == ignore_index);

  const int vecV = V / 8;
  const shortx8* lv = reinterpret_cast<const shortx8*>(lr);
  shortx8* dv = reinterpret_cast<shortx8*>(dr);
  for (int i = threadIdx.x; i < vecV; i += BLOCK) {
    shortx8 v = lv[i], o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (ignored) { o[j] = 0; continue; }
      const long col = (long)i * 8 + j;
      float p = __expf(bf16_to_f32((unsigned short)v[j]) - l);
      float gv = scale * (p - (col == tgt ? 1.f : 0.f));
      o[j] = (short)f32_to_bf16(gv);
    }
    dv[i] = o;
  }
  for (int i = vecV * 8 + threadIdx.x; i < V; i += BLOCK) {
    if (ignored) { dr[i] = 0; continue; }
    float p = __expf(bf16_to_f32(lr[i]) - l);
    dr[i] = f32_to_bf16(scale * (p - ((long)i == tgt ? 1.f : 0.f)));
  }
}

}  // namespace

std::vector<torch::Tensor> cross_entropy_fwd(torch::Tensor logits,
                                             torch::Tensor targets,
                                             long ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kBFloat16
              && logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(targets.dtype() == torch::kInt64 && targets.is_contiguous());
  const long N = logits.size(0);
  const int V = logits.size(1);
  auto losses = torch::empty({N}, logits.options().dtype(torch::kFloat32));
  auto lse = torch::empty({N}, logits.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(N), dim3(BLOCK), 0, stream,
                     (const unsigned short*)logits.data_ptr(),
                     targets.data_ptr<long>(), losses.data_ptr<float>(),
                     lse.data_ptr<float>(), V, ignore_index);
  HIP_CHECK_KERNEL();
  return {losses, lse};
}

torch::Tensor cross_entropy_bwd(torch::Tensor logits, torch::Tensor targets,
                                torch::Tensor lse, torch::Tensor scale,
                                long ignore_index) {
  const long N = logits.size(0);
  const int V = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  // scale stays on device (a .item() here is a host sync that would break
  // hipGraph capture of the training step)
  auto scale_dev = scale.to(logits.device(), torch::kFloat32).contiguous();
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(N), dim3(BLOCK), 0, stream,
                     (const unsigned short*)logits.data_ptr(),
                     targets.data_ptr<long>(), lse.data_ptr<float>(),
                     (unsigned short*)dlogits.data_ptr(),
                     scale_dev.data_ptr<float>(), V, ignore_index);
  HIP_CHECK_KERNEL();
  return dlogits;
}
