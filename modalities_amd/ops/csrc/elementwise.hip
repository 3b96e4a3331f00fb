// K4 RoPE + K6 SiLU-mul: memory-bound elementwise kernels, bf16 vectorized.
// RoPE uses HOST-precomputed cos/sin tables (guide Appendix B: on-device
// trig turns these VALU-bound). Replaces reference RotaryTransform
// (gpt2_model.py:114-229) and SwiGLU's eager silu*mul (model.py:141-157).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

// ---------------- RoPE (rotate-half) --------------------------------------
// x: [B, T, H, D] bf16 contiguous; cos/sin: [T, D/2] f32.
// out_j       = x_j * c_j - x_{j+D/2} * s_j          (j < D/2)
// out_{j+D/2} = x_{j+D/2} * c_j + x_j * s_j
// backward = rotation by -theta (neg=true flips the sin sign).
// One thread handles 2 adjacent (lo, hi) pairs -> 2x ushort2 loads per half.
// in/out may be SLICES of a wider row (fused-QKV layout [B,T,Cq+Ck+Cv]):
// ipitch/opitch are elements between consecutive (b,t) rows, ioff/ooff the
// column offset of this slice's head block. Contiguous [B,T,H,D] tensors
// pass pitch=Hn*D, off=0.
__global__ void rope_kernel(const unsigned short* __restrict__ x,
                            const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t,
                            unsigned short* __restrict__ out,
                            int T, int Hn, int D, int neg,
                            long ipitch, long ioff, long opitch, long ooff) {
  // grid = (ceil(Hn*pairs2 / block), T, B): t and b come straight from
  // blockIdx (no 64-bit div/mod per element — that serialized the old
  // grid-stride form well below HBM bandwidth).
  const int halfD = D / 2;
  const int pairs2 = halfD / 2;
  const int w = blockIdx.x * blockDim.x + threadIdx.x;
  if (w >= Hn * pairs2) return;
  const int h = w / pairs2;
  const int j2 = (w % pairs2) * 2;
  const int t = blockIdx.y;
  const long bt = (long)blockIdx.z * T + t;
  const long base = bt * ipitch + ioff + (long)h * D;
  const long obase = bt * opitch + ooff + (long)h * D;

  const unsigned int lo_u =
      *reinterpret_cast<const unsigned int*>(x + base + j2);
  const unsigned int hi_u =
      *reinterpret_cast<const unsigned int*>(x + base + halfD + j2);
  const float lo0 = bf16_to_f32((unsigned short)(lo_u & 0xffff));
  const float lo1 = bf16_to_f32((unsigned short)(lo_u >> 16));
  const float hi0 = bf16_to_f32((unsigned short)(hi_u & 0xffff));
  const float hi1 = bf16_to_f32((unsigned short)(hi_u >> 16));

  const float c0 = cos_t[t * halfD + j2], c1 = cos_t[t * halfD + j2 + 1];
  float s0 = sin_t[t * halfD + j2], s1 = sin_t[t * halfD + j2 + 1];
  if (neg) { s0 = -s0; s1 = -s1; }

  const float o_lo0 = lo0 * c0 - hi0 * s0;
  const float o_lo1 = lo1 * c1 - hi1 * s1;
  const float o_hi0 = hi0 * c0 + lo0 * s0;
  const float o_hi1 = hi1 * c1 + lo1 * s1;

  *reinterpret_cast<unsigned int*>(out + obase + j2) =
      (unsigned int)f32_to_bf16(o_lo0) | ((unsigned int)f32_to_bf16(o_lo1) << 16);
  *reinterpret_cast<unsigned int*>(out + obase + halfD + j2) =
      (unsigned int)f32_to_bf16(o_hi0) | ((unsigned int)f32_to_bf16(o_hi1) << 16);
}

// ---------------- SiLU * mul ----------------------------------------------
__global__ void silu_mul_fwd_kernel(const unsigned short* __restrict__ g,
                                    const unsigned short* __restrict__ u,
                                    unsigned short* __restrict__ out, long n8) {
  const shortx8* gv = reinterpret_cast<const shortx8*>(g);
  const shortx8* uv = reinterpret_cast<const shortx8*>(u);
  shortx8* ov = reinterpret_cast<shortx8*>(out);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    shortx8 gg = gv[i], uu = uv[i], oo;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf16_to_f32((unsigned short)gg[j]);
      float uf = bf16_to_f32((unsigned short)uu[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      oo[j] = (short)f32_to_bf16(gf * sig * uf);
    }
    ov[i] = oo;
  }
}

__global__ void silu_mul_bwd_kernel(const unsigned short* __restrict__ dout,
                                    const unsigned short* __restrict__ g,
                                    const unsigned short* __restrict__ u,
                                    unsigned short* __restrict__ dg,
                                    unsigned short* __restrict__ du, long n8) {
  const shortx8* dv = reinterpret_cast<const shortx8*>(dout);
  const shortx8* gv = reinterpret_cast<const shortx8*>(g);
  const shortx8* uv = reinterpret_cast<const shortx8*>(u);
  shortx8* dgv = reinterpret_cast<shortx8*>(dg);
  shortx8* duv = reinterpret_cast<shortx8*>(du);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    shortx8 dd = dv[i], gg = gv[i], uu = uv[i], og, ou;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float df = bf16_to_f32((unsigned short)dd[j]);
      float gf = bf16_to_f32((unsigned short)gg[j]);
      float uf = bf16_to_f32((unsigned short)uu[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      float silu = gf * sig;
      float dsilu = sig * (1.f + gf * (1.f - sig));
      og[j] = (short)f32_to_bf16(df * uf * dsilu);
      ou[j] = (short)f32_to_bf16(df * silu);
    }
    dgv[i] = og;
    duv[i] = ou;
  }
}

// Joint-layout variant: the gate/up halves live in ONE [rows, 2H] tensor
// (the packed SwiGLU up-projection output: a single GEMM instead of two).
// fwd reads both halves of a row; bwd writes the joint gradient buffer
// directly (no cat before the packed GEMM backward).
__global__ void silu_mul_joint_fwd_kernel(const unsigned short* __restrict__ wv,
                                          unsigned short* __restrict__ out,
                                          long n8, long h8) {
  shortx8* ov = reinterpret_cast<shortx8*>(out);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    const long row = i / h8, c8 = i % h8;
    const shortx8 gg = *reinterpret_cast<const shortx8*>(
        wv + (row * 2 * h8 + c8) * 8);
    const shortx8 uu = *reinterpret_cast<const shortx8*>(
        wv + (row * 2 * h8 + h8 + c8) * 8);
    shortx8 oo;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf16_to_f32((unsigned short)gg[j]);
      float uf = bf16_to_f32((unsigned short)uu[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      oo[j] = (short)f32_to_bf16(gf * sig * uf);
    }
    ov[i] = oo;
  }
}

__global__ void silu_mul_joint_bwd_kernel(const unsigned short* __restrict__ dout,
                                          const unsigned short* __restrict__ wv,
                                          unsigned short* __restrict__ dwv,
                                          long n8, long h8) {
  const shortx8* dv = reinterpret_cast<const shortx8*>(dout);
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    const long row = i / h8, c8 = i % h8;
    const long gbase = (row * 2 * h8 + c8) * 8;
    const long ubase = (row * 2 * h8 + h8 + c8) * 8;
    const shortx8 gg = *reinterpret_cast<const shortx8*>(wv + gbase);
    const shortx8 uu = *reinterpret_cast<const shortx8*>(wv + ubase);
    shortx8 dd = dv[i], og, ou;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float df = bf16_to_f32((unsigned short)dd[j]);
      float gf = bf16_to_f32((unsigned short)gg[j]);
      float uf = bf16_to_f32((unsigned short)uu[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      float silu = gf * sig;
      float dsilu = sig * (1.f + gf * (1.f - sig));
      og[j] = (short)f32_to_bf16(df * uf * dsilu);
      ou[j] = (short)f32_to_bf16(df * silu);
    }
    *reinterpret_cast<shortx8*>(dwv + gbase) = og;
    *reinterpret_cast<shortx8*>(dwv + ubase) = ou;
  }
}

int grid_for(long work, int block) {
  long g = (work + block - 1) / block;
  return (int)min(g, (long)(256 * 8));  // cap + grid-stride (guide G11)
}

}  // namespace

torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cos_t, torch::Tensor sin_t,
                       bool neg) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.dim() == 4
              && x.is_contiguous(), "rope: x must be contiguous bf16 [B,T,H,D]");
  const int B = x.size(0), T = x.size(1), Hn = x.size(2), D = x.size(3);
  TORCH_CHECK(D % 4 == 0, "rope: head_dim must be divisible by 4");
  TORCH_CHECK(T <= 65535 && cos_t.size(0) >= T && cos_t.size(1) == D / 2);
  TORCH_CHECK(cos_t.dtype() == torch::kFloat32 && cos_t.is_contiguous());
  auto out = torch::empty_like(x);
  auto stream = at::cuda::getCurrentHIPStream();
  const int work = Hn * (D / 4);
  hipLaunchKernelGGL(rope_kernel, dim3((work + 255) / 256, T, B), dim3(256),
                     0, stream, (const unsigned short*)x.data_ptr(),
                     cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
                     (unsigned short*)out.data_ptr(), T, Hn, D, neg ? 1 : 0,
                     (long)Hn * D, 0L, (long)Hn * D, 0L);
  HIP_CHECK_KERNEL();
  return out;
}

// RoPE a head-block slice of a fused-QKV activation [B, T, Ctot] into a
// contiguous [B, T, Hn, D] tensor (no .contiguous() copy of the slice).
torch::Tensor rope_fwd_slice(torch::Tensor qkv, torch::Tensor cos_t,
                             torch::Tensor sin_t, long col_off, long Hn,
                             long D, bool neg) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == torch::kBFloat16
              && qkv.dim() == 3 && qkv.is_contiguous());
  const int B = qkv.size(0), T = qkv.size(1);
  const long Ctot = qkv.size(2);
  TORCH_CHECK(col_off % 2 == 0 && (D % 4) == 0 && col_off + Hn * D <= Ctot);
  TORCH_CHECK(T <= 65535 && cos_t.size(0) >= T && cos_t.size(1) == D / 2
              && cos_t.dtype() == torch::kFloat32 && cos_t.is_contiguous());
  auto out = torch::empty({B, T, Hn, D}, qkv.options());
  auto stream = at::cuda::getCurrentHIPStream();
  const int work = (int)(Hn * (D / 4));
  hipLaunchKernelGGL(rope_kernel, dim3((work + 255) / 256, T, B), dim3(256),
                     0, stream, (const unsigned short*)qkv.data_ptr(),
                     cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
                     (unsigned short*)out.data_ptr(), T, (int)Hn,
                     (int)D, neg ? 1 : 0, Ctot, col_off, Hn * D, 0L);
  HIP_CHECK_KERNEL();
  return out;
}

// Inverse-RoPE a contiguous [B, T, Hn, D] gradient into a head-block slice
// of a fused-QKV gradient buffer [B, T, Ctot] (replaces the backward cat).
void rope_bwd_slice(torch::Tensor dx, torch::Tensor cos_t, torch::Tensor sin_t,
                    torch::Tensor dqkv, long col_off) {
  TORCH_CHECK(dx.is_cuda() && dx.dtype() == torch::kBFloat16 && dx.dim() == 4
              && dx.is_contiguous());
  TORCH_CHECK(dqkv.is_cuda() && dqkv.dtype() == torch::kBFloat16
              && dqkv.dim() == 3 && dqkv.is_contiguous());
  const int B = dx.size(0), T = dx.size(1), Hn = dx.size(2), D = dx.size(3);
  const long Ctot = dqkv.size(2);
  TORCH_CHECK(dqkv.size(0) == B && dqkv.size(1) == T
              && col_off + (long)Hn * D <= Ctot && col_off % 2 == 0);
  TORCH_CHECK(T <= 65535 && cos_t.size(0) >= T && cos_t.size(1) == D / 2);
  auto stream = at::cuda::getCurrentHIPStream();
  const int work = Hn * (D / 4);
  hipLaunchKernelGGL(rope_kernel, dim3((work + 255) / 256, T, B), dim3(256),
                     0, stream, (const unsigned short*)dx.data_ptr(),
                     cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
                     (unsigned short*)dqkv.data_ptr(), T, Hn, D, 1,
                     (long)Hn * D, 0L, Ctot, col_off);
  HIP_CHECK_KERNEL();
}

torch::Tensor silu_mul_fwd(torch::Tensor g, torch::Tensor u) {
  TORCH_CHECK(g.is_cuda() && g.dtype() == torch::kBFloat16 && g.is_contiguous());
  TORCH_CHECK(g.numel() % 8 == 0, "silu_mul: numel must be divisible by 8");
  auto out = torch::empty_like(g);
  long n8 = g.numel() / 8;
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(silu_mul_fwd_kernel, dim3(grid_for(n8, 256)), dim3(256), 0,
                     stream, (const unsigned short*)g.data_ptr(),
                     (const unsigned short*)u.data_ptr(),
                     (unsigned short*)out.data_ptr(), n8);
  HIP_CHECK_KERNEL();
  return out;
}

std::vector<torch::Tensor> silu_mul_bwd(torch::Tensor dout, torch::Tensor g,
                                        torch::Tensor u) {
  auto dg = torch::empty_like(g);
  auto du = torch::empty_like(u);
  long n8 = g.numel() / 8;
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(silu_mul_bwd_kernel, dim3(grid_for(n8, 256)), dim3(256), 0,
                     stream, (const unsigned short*)dout.data_ptr(),
                     (const unsigned short*)g.data_ptr(),
                     (const unsigned short*)u.data_ptr(),
                     (unsigned short*)dg.data_ptr(),
                     (unsigned short*)du.data_ptr(), n8);
  HIP_CHECK_KERNEL();
  return {dg, du};
}


torch::Tensor silu_mul_joint_fwd(torch::Tensor wv) {
  TORCH_CHECK(wv.is_cuda() && wv.dtype() == torch::kBFloat16 && wv.is_contiguous());
  const long H = wv.size(-1) / 2;
  TORCH_CHECK(wv.size(-1) % 2 == 0 && H % 8 == 0,
              "silu_mul_joint: last dim must be 2*H with H %% 8 == 0");
  auto sizes = wv.sizes().vec();
  sizes.back() = H;
  auto out = torch::empty(sizes, wv.options());
  const long rows = wv.numel() / (2 * H);
  const long n8 = rows * (H / 8);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(silu_mul_joint_fwd_kernel, dim3(grid_for(n8, 256)),
                     dim3(256), 0, stream,
                     (const unsigned short*)wv.data_ptr(),
                     (unsigned short*)out.data_ptr(), n8, H / 8);
  HIP_CHECK_KERNEL();
  return out;
}

torch::Tensor silu_mul_joint_bwd(torch::Tensor dout, torch::Tensor wv) {
  TORCH_CHECK(dout.is_cuda() && dout.is_contiguous() && wv.is_contiguous());
  const long H = wv.size(-1) / 2;
  auto dwv = torch::empty_like(wv);
  const long rows = wv.numel() / (2 * H);
  const long n8 = rows * (H / 8);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(silu_mul_joint_bwd_kernel, dim3(grid_for(n8, 256)),
                     dim3(256), 0, stream,
                     (const unsigned short*)dout.data_ptr(),
                     (const unsigned short*)wv.data_ptr(),
                     (unsigned short*)dwv.data_ptr(), n8, H / 8);
  HIP_CHECK_KERNEL();
  return dwv;
}
