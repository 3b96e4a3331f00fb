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
template <int SLOTS>
__global__ void rmsnorm_bwd_kernel(const unsigned short* __restrict__ dy,
                                   const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ w,
                                   const float* __restrict__ invrms,
                                   unsigned short* __restrict__ dx,
                                   float* __restrict__ dw,  // f32 accumulator
                                   int H, long N, int rows_per_block) {
  // One WAVE per row, and dw contributions accumulate in REGISTERS: lane l
  // always touches vec8 columns {l, l+64, ...} (SLOTS of them), so its dw
  // partials live in dwacc[][] across the whole row loop and spill to the
  // per-wave LDS slab exactly once. (v1 block-per-row paid a barrier per
  // row; v2 wave-per-row paid 16 LDS read-modify-writes per 16B of HBM
  // traffic — both measured well under 20% of HBM bandwidth.)
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* slab = reinterpret_cast<float*>(smem_raw);
  const int nwaves = blockDim.x / WAVE_SIZE;
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  float* my_slab = slab + (long)wid * H;

  const long row0 = (long)blockIdx.x * rows_per_block;
  const long row1 = min(row0 + rows_per_block, N);
  const int vecH = H / 8;

  float dwacc[SLOTS][8];
#pragma unroll
  for (int it = 0; it < SLOTS; ++it)
#pragma unroll
    for (int j = 0; j < 8; ++j) dwacc[it][j] = 0.f;

  const shortx8* wv = reinterpret_cast<const shortx8*>(w);
  for (long row = row0 + wid; row < row1; row += nwaves) {
    const shortx8* xv = reinterpret_cast<const shortx8*>(x + row * (long)H);
    const shortx8* dyv = reinterpret_cast<const shortx8*>(dy + row * (long)H);
    shortx8* dxv = reinterpret_cast<shortx8*>(dx + row * (long)H);
    const float r = invrms[row];

    float dot = 0.f;
#pragma unroll
    for (int it = 0; it < SLOTS; ++it) {
      const int i = lane + it * WAVE_SIZE;
      if (i < vecH) {
        shortx8 xvv = xv[i], dyvv = dyv[i], wvv = wv[i];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          dot += bf16_to_f32((unsigned short)dyvv[j]) *
                 bf16_to_f32((unsigned short)wvv[j]) *
                 bf16_to_f32((unsigned short)xvv[j]);
        }
      }
    }
    dot = wave_reduce_sum(dot);
    const float k = r * r * r * dot / H;

#pragma unroll
    for (int it = 0; it < SLOTS; ++it) {
      const int i = lane + it * WAVE_SIZE;
      if (i < vecH) {
        shortx8 xvv = xv[i], dyvv = dyv[i], wvv = wv[i], o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xf = bf16_to_f32((unsigned short)xvv[j]);
          float dyf = bf16_to_f32((unsigned short)dyvv[j]);
          float wf = bf16_to_f32((unsigned short)wvv[j]);
          o[j] = (short)f32_to_bf16(r * wf * dyf - k * xf);
          dwacc[it][j] += dyf * xf * r;
        }
        dxv[i] = o;
      }
    }
  }

#pragma unroll
  for (int it = 0; it < SLOTS; ++it) {
    const int i = lane + it * WAVE_SIZE;
    if (i < vecH)
#pragma unroll
      for (int j = 0; j < 8; ++j) my_slab[i * 8 + j] = dwacc[it][j];
  }
  __syncthreads();
  // Per-block partial row into global scratch (dw here = dw_part, laid out
  // [nblocks, H]). A direct atomicAdd merge serializes nblocks (~2048)
  // fp32 atomics per address — measured as the kernel's dominant cost —
  // and is nondeterministic; the two-phase reduce is neither.
  for (int i = threadIdx.x; i < H; i += blockDim.x) {
    float acc = 0.f;
    for (int sl = 0; sl < nwaves; ++sl) acc += slab[(long)sl * H + i];
    dw[(long)blockIdx.x * H + i] = acc;
  }
}

// Column reduce of dw_part [nblocks, H] -> dw [H]. grid.y chunks the block
// dimension; the few cross-chunk merges go through atomicAdd (into zeros).
__global__ void dw_reduce_kernel(const float* __restrict__ dw_part,
                                 float* __restrict__ dw, int H, int nblocks,
                                 int rows_per_chunk) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= H) return;
  const int b0 = blockIdx.y * rows_per_chunk;
  const int b1 = min(b0 + rows_per_chunk, nblocks);
  float acc = 0.f;
  for (int b = b0; b < b1; ++b) acc += dw_part[(long)b * H + i];
  if (gridDim.y == 1) dw[i] = acc;
  else atomicAdd(&dw[i], acc);
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
  torch::Tensor dw_part;  // allocated after nblocks is known
  // Grid-stride over rows: cap blocks so the LDS dw accumulator amortizes.
  int nblocks = (int)min((long)2048, (N + 3) / 4);
  int rows_per_block = (int)((N + nblocks - 1) / nblocks);
  nblocks = (int)((N + rows_per_block - 1) / rows_per_block);
  const int nwaves = BLOCK / WAVE_SIZE;
  size_t smem = (size_t)H * nwaves * sizeof(float);
  TORCH_CHECK(smem <= 160 * 1024, "rmsnorm_bwd: H too large for LDS slabs");
  TORCH_CHECK(H % 8 == 0, "rmsnorm_bwd: H must be divisible by 8");
  const int slots = (H / 8 + WAVE_SIZE - 1) / WAVE_SIZE;
  dw_part = torch::empty({(long)nblocks, (long)H},
                         x.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentHIPStream();
  auto launch = [&](auto* kfn) {
    hipLaunchKernelGGL(kfn, dim3(nblocks), dim3(BLOCK), smem, stream,
                       (const unsigned short*)dy.data_ptr(),
                       (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       invrms.data_ptr<float>(),
                       (unsigned short*)dx.data_ptr(),
                       dw_part.data_ptr<float>(), H, N, rows_per_block);
  };
  if (slots <= 1) launch(rmsnorm_bwd_kernel<1>);
  else if (slots <= 2) launch(rmsnorm_bwd_kernel<2>);
  else if (slots <= 5) launch(rmsnorm_bwd_kernel<5>);
  else if (slots <= 8) launch(rmsnorm_bwd_kernel<8>);
  else if (slots <= 16) launch(rmsnorm_bwd_kernel<16>);
  else TORCH_CHECK(false, "rmsnorm_bwd: H too large (max 8192)");
  {
    const int chunks = (int)min((long)32, (long)((nblocks + 63) / 64));
    const int rpc = (nblocks + chunks - 1) / chunks;
    hipLaunchKernelGGL(dw_reduce_kernel,
                       dim3((H + BLOCK - 1) / BLOCK, chunks), dim3(BLOCK), 0,
                       stream, dw_part.data_ptr<float>(),
                       dw.data_ptr<float>(), H, nblocks, rpc);
  }
  HIP_CHECK_KERNEL();
  return {dx, dw};
}
