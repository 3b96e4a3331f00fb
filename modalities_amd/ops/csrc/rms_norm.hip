// K5: RMSNorm fwd/bwd for bf16 activations, f32 accumulation.
// One 256-thread workgroup per row; bf16x8 vectorized loads (guide G13:
// scalar bf16 loads are ~2x slower). Memory-bound: target HBM roofline.
// Replaces the reference's nn.RMSNorm (reference: models/components/
// layer_norms.py:9-65).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr int BLOCK = 256;

// -------- forward: y = x * rsqrt(mean(x^2)+eps) * w;  saves invrms --------
__global__ void rmsnorm_fwd_kernel(const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ w,
                                   unsigned short* __restrict__ y,
                                   float* __restrict__ invrms,
                                   int H, float eps) {
  __shared__ float scratch[BLOCK / WAVE_SIZE];
  const long row = blockIdx.x;
  const unsigned short* xr = x + row * (long)H;
  unsigned short* yr = y + row * (long)H;

  float ss = 0.f;
  const int vecH = H / 8;
  const shortx8* xv = reinterpret_cast<const shortx8*>(xr);
  for (int i = threadIdx.x; i < vecH; i += BLOCK) {
    shortx8 v = xv[i];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32((unsigned short)v[j]);
      ss += f * f;
    }
  }
  for (int i = vecH * 8 + threadIdx.x; i < H; i += BLOCK) {
    float f = bf16_to_f32(xr[i]);
    ss += f * f;
  }
  ss = block_reduce_sum(ss, scratch);
  const float r = rsqrtf(ss / H + eps);
  if (threadIdx.x == 0) invrms[row] = r;

  const shortx8* wv = reinterpret_cast<const shortx8*>(w);
  shortx8* yv = reinterpret_cast<shortx8*>(yr);
  for (int i = threadIdx.x; i < vecH; i += BLOCK) {
    shortx8 xvv = xv[i], wvv = wv[i], o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32((unsigned short)xvv[j]) * r *
                bf16_to_f32((unsigned short)wvv[j]);
      o[j] = (short)f32_to_bf16(f);
    }
    yv[i] = o;
  }
  for (int i = vecH * 8 + threadIdx.x; i < H; i += BLOCK) {
    yr[i] = f32_to_bf16(bf16_to_f32(xr[i]) * r * bf16_to_f32(w[i]));
  }
}

// -------- backward ---------------------------------------------------------
// dx_j = r*w_j*dy_j - r^3 * x_j / H * sum_i(dy_i * w_i * x_i)
// dw_j = sum_rows dy_j * x_j * r   (accumulated per-block in LDS, one
// atomicAdd per element per block — guide G12 contention rule).
__global__ void rmsnorm_bwd_kernel(const unsigned short* __restrict__ dy,
                                   const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ w,
                                   const float* __restrict__ invrms,
                                   unsigned short* __restrict__ dx,
                                   float* __restrict__ dw,  // f32 accumulator
                                   int H, long N, int rows_per_block) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* dw_local = reinterpret_cast<float*>(smem_raw);            // H floats
  float* scratch = dw_local + H;                                   // BLOCK/64

  for (int i = threadIdx.x; i < H; i += BLOCK) dw_local[i] = 0.f;
  __syncthreads();

  const long row0 = (long)blockIdx.x * rows_per_block;
  const long row1 = min(row0 + rows_per_block, N);
  const int vecH = H / 8;

  for (long row = row0; row < row1; ++row) {
    const unsigned short* xr = x + row * (long)H;
    const unsigned short* dyr = dy + row * (long)H;
    unsigned short* dxr = dx + row * (long)H;
    const float r = invrms[row];

    const shortx8* xv = reinterpret_cast<const shortx8*>(xr);
    const shortx8* dyv = reinterpret_cast<const shortx8*>(dyr);
    const shortx8* wv = reinterpret_cast<const shortx8*>(w);

    float dot = 0.f;
    for (int i = threadIdx.x; i < vecH; i += BLOCK) {
      shortx8 xvv = xv[i], dyvv = dyv[i], wvv = wv[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        dot += bf16_to_f32((unsigned short)dyvv[j]) *
               bf16_to_f32((unsigned short)wvv[j]) *
               bf16_to_f32((unsigned short)xvv[j]);
      }
    }
    for (int i = vecH * 8 + threadIdx.x; i < H; i += BLOCK)
      dot += bf16_to_f32(dyr[i]) * bf16_to_f32(w[i]) * bf16_to_f32(xr[i]);
    dot = block_reduce_sum(dot, scratch);
    const float k = r * r * r * dot / H;

    shortx8* dxv = reinterpret_cast<shortx8*>(dxr);
    for (int i = threadIdx.x; i < vecH; i += BLOCK) {
      shortx8 xvv = xv[i], dyvv = dyv[i], wvv = wv[i], o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xf = bf16_to_f32((unsigned short)xvv[j]);
        float dyf = bf16_to_f32((unsigned short)dyvv[j]);
        float wf = bf16_to_f32((unsigned short)wvv[j]);
        o[j] = (short)f32_to_bf16(r * wf * dyf - k * xf);
        dw_local[i * 8 + j] += dyf * xf * r;
      }
      dxv[i] = o;
    }
    for (int i = vecH * 8 + threadIdx.x; i < H; i += BLOCK) {
      float xf = bf16_to_f32(xr[i]), dyf = bf16_to_f32(dyr[i]),
            wf = bf16_to_f32(w[i]);
      dxr[i] = f32_to_bf16(r * wf * dyf - k * xf);
      dw_local[i] += dyf * xf * r;
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < H; i += BLOCK) {
    if (dw_local[i] != 0.f) atomicAdd(&dw[i], dw_local[i]);
  }
}

}  // namespace

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16, "x must be device bf16");
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(w.dtype() == torch::kBFloat16 && w.is_contiguous());
  const long N = x.size(0);
  const int H = x.size(1);
  auto y = torch::empty_like(x);
  auto invrms = torch::empty({N}, x.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3(N), dim3(BLOCK), 0, stream,
                     (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)w.data_ptr(),
                     (unsigned short*)y.data_ptr(),
                     invrms.data_ptr<float>(), H, (float)eps);
  HIP_CHECK_KERNEL();
  return {y, invrms};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor invrms) {
  const long N = x.size(0);
  const int H = x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({H}, x.options().dtype(torch::kFloat32));
  // Grid-stride over rows: cap blocks so the LDS dw accumulator amortizes.
  int nblocks = (int)min((long)2048, N);
  int rows_per_block = (int)((N + nblocks - 1) / nblocks);
  nblocks = (int)((N + rows_per_block - 1) / rows_per_block);
  size_t smem = (H + BLOCK / WAVE_SIZE) * sizeof(float);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(rmsnorm_bwd_kernel, dim3(nblocks), dim3(BLOCK), smem, stream,
                     (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)w.data_ptr(),
                     invrms.data_ptr<float>(),
                     (unsigned short*)dx.data_ptr(),
                     dw.data_ptr<float>(), H, N, rows_per_block);
  HIP_CHECK_KERNEL();
  return {dx, dw};
}
