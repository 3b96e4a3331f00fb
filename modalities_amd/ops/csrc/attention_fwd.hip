// K1 (forward): flash-style causal attention for CDNA4/gfx950, GQA-aware.
//
// Replaces the reference's flash-attn wheel / torch SDPA paths (reference:
// src/modalities/models/gpt2/gpt2_model.py:595-658) with a hand-written
// MFMA kernel. Structure follows the CDNA4 guide's attention ladder:
//
// - workgroup = 4 waves; each wave owns a 32-row Q block (WG covers 128 rows)
// - KV tiles of 64 staged in LDS by the whole WG:
//     K  [64][D]          row-major, XOR-swizzled (guide T2)
//     V  [D/16][64][16]   panel image read via ds_read_b64_tr_b16 (guide
//                         T10 hardware transpose; staged with 16B writes)
// - SWAPPED QK^T: S^T[kv][q] = mfma(A=K, B=Q^T) puts a full softmax row in
//   one lane pair (q = lane&31) -> row reduce is in-register + 1 shfl_xor(32)
// - SWAPPED PV:   O^T[d][q]  = mfma(A=V^T, B=P^T): the online-softmax O
//   rescale stays lane-local (same q = lane&31 layout as the stats)
// - P (f32, S-layout) -> PV A-fragment (bf16) via cvt_pk + permlane32_swap
//   (guide T12) — no LDS round-trip for P.
// - defer-max rescale skip (guide T13, THR=8) and exp2-domain softmax
//   state (log2e folded into the QK scale).
//
// D in {64, 128}; causal only. bf16 in/out, f32 softmax state; lse saved
// for the backward (natural-log domain).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(2))) unsigned int uintx2v;

namespace attn {

// Hardware transpose read (semantics verified by tr16_probe; see the
// derivation note in attention_bwd.hip): from a [panel][row][16-col] LDS
// image, lane l receives X[k0+j][d0+(l&15)] — the MFMA A/B k-run.
__device__ __forceinline__ uintx2v tr_read_b64(unsigned addr_bytes) {
  uintx2v r;
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(r) : "v"(addr_bytes));
  return r;
}

__device__ __forceinline__ unsigned lds_addr(const void* p) {
  return (unsigned)(unsigned long long)p;
}

constexpr int QBLK = 32;     // q rows per wave
constexpr int NWAVES = 4;    // waves per workgroup
constexpr int WG_Q = QBLK * NWAVES;  // 128 q rows per workgroup
constexpr int KVBLK = 64;    // kv rows per LDS tile

// Row-major tile swizzle (read rows differ within {0..3,16..19}+32k b128
// lane groups): key distinct over that set -> conflict-free reads; staging
// writes are 8-lane same-row groups with distinct 16B slots -> conflict-free.
__device__ __forceinline__ int swz(int row, int col) {
  return col ^ (((row & 3) | (((row >> 4) & 1) << 2)) << 3);
}

// C/D register map of v_mfma_f32_32x32x16_bf16 (guide §3):
//   col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
__device__ __forceinline__ int crow(int reg, int hi) {
  return (reg & 3) + 8 * (reg >> 2) + 4 * hi;
}

// A/B fragment k-index for 32x32x16: lane provides elements k = hi*8 + m.
// ABL (ablation, guide mistake #8 "ablate before optimizing"):
//   0 = full kernel; 1 = skip V-transpose staging; 2 = skip all staging;
//   3 = skip softmax VALU (P := raw S). Modes 1-3 give WRONG results and
//   exist only for cost attribution.
template <int D_, int ABL = 0>
struct AttnFwdKernel {
  static constexpr int D = D_;
  static constexpr int NDSTEP = D / 16;   // QK^T contraction steps
  static constexpr int NDBLK = D / 32;    // PV output column blocks

  // LDS: K tile + V^T tile (both swizzled, bf16)
  struct Smem {
    unsigned short k[KVBLK * D];
    unsigned short vt[D * KVBLK];
  };

  static __device__ void run(const unsigned short* __restrict__ q,
                             const unsigned short* __restrict__ k,
                             const unsigned short* __restrict__ v,
                             unsigned short* __restrict__ o,
                             float* __restrict__ lse,
                             int B, int Tq, int Tkv, int q_off, int Hq,
                             int Hkv, float scale, char* smem_raw) {
    Smem* sm = reinterpret_cast<Smem*>(smem_raw);
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int hi = lane >> 5;
    const int ln31 = lane & 31;

    const int qblk0 = blockIdx.x * WG_Q;       // first q row of workgroup
    const int h = blockIdx.y;
    const int b = blockIdx.z;
    const int hkv = h / (Hq / Hkv);

    const long q_base = (((long)b * Tq) * Hq + h) * D;       // + t*Hq*D
    const long kv_base = (((long)b * Tkv) * Hkv + hkv) * D;
    const int qg = qblk0 + wid * QBLK + ln31;  // this lane's LOCAL q row
    const int qgl = qg + q_off;                // global position vs keys
    // exp2 fold: v_exp_f32 is base-2; folding log2(e) into the QK scale and
    // keeping the softmax state in the log2 domain removes one VALU multiply
    // per score element. lse is converted back to natural log on store.
    const float scale2 = scale * 1.44269504f;

    // ---- Q preload: lane holds Q[qg][hi*8 + m + 16*s], s=0..NDSTEP-1 ----
    bf16x8 qfrag[NDSTEP];
    {
      const unsigned short* qr = q + q_base + (long)qg * Hq * D;
#pragma unroll
      for (int s = 0; s < NDSTEP; ++s) {
        if (qg < Tq) {
          qfrag[s] = *reinterpret_cast<const bf16x8*>(qr + hi * 8 + 16 * s);
        } else {
          for (int m = 0; m < 8; ++m) qfrag[s][m] = (__bf16)0.f;
        }
      }
    }

    // ---- softmax state (per lane: q row = ln31; both hi halves track) ----
    float m_run = -INFINITY, l_run = 0.f;
    floatx16 acc_o[NDBLK];
#pragma unroll
    for (int dblk = 0; dblk < NDBLK; ++dblk)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc_o[dblk][r] = 0.f;

    // causal upper bound in GLOBAL key coords
    const int q_hi_wg = min(qblk0 + WG_Q - 1, Tq - 1) + q_off;
    const int n_tiles = (min(q_hi_wg, Tkv - 1)) / KVBLK + 1;

    // V goes into a panel image [D/16][KVBLK][16] consumed via hardware
    // transpose reads (vectorized writes; the v1 scalar scatter stores had
    // 8-way write-bank conflicts). Staging stays single-buffered in-loop:
    // the T14 issue-early variant costs 32 VGPRs and drops this kernel
    // from 2 waves/SIMD to 1, which measured slower.
    constexpr int NCH = KVBLK * D / 8 / (NWAVES * 64);
    const int tid = threadIdx.x;

    for (int tile = 0; tile < n_tiles; ++tile) {
      const int kv0 = tile * KVBLK;
#pragma unroll
      for (int it = 0; it < NCH; ++it) {
        const int i = tid + it * NWAVES * 64;
        const int r = (i * 8) / D, c = (i * 8) % D;
        shortx8 kk, vv;
        if (kv0 + r < Tkv) {
          kk = *reinterpret_cast<const shortx8*>(
              k + kv_base + (long)(kv0 + r) * Hkv * D + c);
          vv = *reinterpret_cast<const shortx8*>(
              v + kv_base + (long)(kv0 + r) * Hkv * D + c);
        } else {
#pragma unroll
          for (int m = 0; m < 8; ++m) { kk[m] = 0; vv[m] = 0; }
        }
        if constexpr (ABL < 2)
          *reinterpret_cast<shortx8*>(&sm->k[r * D + swz(r, c)]) = kk;
        if constexpr (ABL < 1) {
          *reinterpret_cast<shortx8*>(
              &sm->vt[(c >> 4) * (KVBLK * 16) + r * 16 + (c & 15)]) = vv;
        } else {
          asm volatile("" :: "v"(vv[0]));  // keep loads alive (rule 17)
        }
      }
      __syncthreads();

      // ---- S^T = K Q^T : two 32x32 accs (kv 0-31, 32-63) ----
      floatx16 s0, s1;
#pragma unroll
      for (int r = 0; r < 16; ++r) { s0[r] = 0.f; s1[r] = 0.f; }
#pragma unroll
      for (int s = 0; s < NDSTEP; ++s) {
        const int col = hi * 8 + 16 * s;
        const int r0 = ln31, r1 = ln31 + 32;
        bf16x8 ka = *reinterpret_cast<const bf16x8*>(&sm->k[r0 * D + swz(r0, col)]);
        bf16x8 kb = *reinterpret_cast<const bf16x8*>(&sm->k[r1 * D + swz(r1, col)]);
        s0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[s], s0, 0, 0, 0);
        s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kb, qfrag[s], s1, 0, 0, 0);
      }

      // ---- scale + causal mask; per-lane P rows (q = ln31) --------------
      float p[32];
      if constexpr (ABL == 3) {  // softmax cost ablation: P := raw S
#pragma unroll
        for (int r = 0; r < 16; ++r) { p[r] = s0[r]; p[16 + r] = s1[r]; }
        l_run += 1.f;
      } else {
      float tmax = -INFINITY;
      // interior fast path: this wave's lowest q row covers the whole kv
      // tile and the tile is in range -> no causal/range compares (the
      // kernel is VALU-issue-bound; dropping 64 cmp+cndmask per tile)
      const int q_lo_wave = qblk0 + wid * QBLK + q_off;
      if (kv0 + KVBLK - 1 <= q_lo_wave && kv0 + KVBLK <= Tkv && qg < Tq) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float v0 = s0[r] * scale2;
          float v1 = s1[r] * scale2;
          p[r] = v0;
          p[16 + r] = v1;
          tmax = fmaxf(tmax, fmaxf(v0, v1));
        }
      } else {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kg0 = kv0 + crow(r, hi);
        const int kg1 = kv0 + 32 + crow(r, hi);
        float v0 = (kg0 <= qgl && kg0 < Tkv) ? s0[r] * scale2 : -INFINITY;
        float v1 = (kg1 <= qgl && kg1 < Tkv) ? s1[r] * scale2 : -INFINITY;
        p[r] = v0;
        p[16 + r] = v1;
        tmax = fmaxf(tmax, fmaxf(v0, v1));
      }
      }
      tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));  // other kv half, same q

      // ---- online softmax update ----------------------------------------
      // defer-max (guide T13): when no lane's tile max exceeds the running
      // max by more than RESCALE_THR, keep the old max — skips the O-wide
      // rescale (NDBLK*16 mults) and the l/alpha work; P is then bounded by
      // exp(THR) instead of 1, which f32 accumulation and bf16 P tolerate
      // (~3x max-abs error vs THR=0; decision taken BEFORE this tile's P is
      // exponentiated, so no pending P*V is split — the T13 hazard order).
      constexpr float RESCALE_THR = 8.0f;
      const bool defer = __all(tmax - m_run <= RESCALE_THR);
      if (!defer) {
        const float m_new = fmaxf(m_run, tmax);
        const float m_safe = (m_new == -INFINITY) ? 0.f : m_new;
        // -inf guards: fully-masked tiles keep O/l at 0 without NaNs
        const float alpha = (m_run == -INFINITY) ? 0.f : __builtin_amdgcn_exp2f(m_run - m_safe);
        l_run *= alpha;
#pragma unroll
        for (int dblk = 0; dblk < NDBLK; ++dblk)
#pragma unroll
          for (int r = 0; r < 16; ++r) acc_o[dblk][r] *= alpha;
        m_run = m_new;
      }
      const float m_safe2 = (m_run == -INFINITY) ? 0.f : m_run;
      float psum = 0.f;
#pragma unroll
      for (int r = 0; r < 32; ++r) {
        p[r] = __builtin_amdgcn_exp2f(p[r] - m_safe2);
        psum += p[r];
      }
      psum += __shfl_xor(psum, 32, 64);
      l_run += psum;
      }  // ABL != 3

      // ---- P (f32, S-layout) -> PV A/B fragment (bf16) -------------------
      // Lane holds P[q=ln31][kv in crow set]. The PV B-operand needs
      // P^T[k=kv = hi*8+m+16*s][n=q=ln31]. Word pairs plus one
      // permlane32_swap per pair produce the fragment (guide T12 analysis):
      //   after swap(w0=cvtpk(p0,p1), w2=cvtpk(p4,p5)):
      //     lanes hi=0: w0 = kv(0,1), w2 = kv(4,5)
      //     lanes hi=1: w0 = kv(8,9), w2 = kv(12,13)
      bf16x8 pfrag[KVBLK / 16];
#pragma unroll
      for (int s = 0; s < KVBLK / 16; ++s) {
        // registers covering kv block s*16..s*16+15 in this lane: p[8s..8s+7]
        unsigned int w[4];
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const float lo = p[8 * s + 2 * i];
          const float hi_v = p[8 * s + 2 * i + 1];
          w[i] = ((unsigned int)f32_to_bf16(lo)) |
                 ((unsigned int)f32_to_bf16(hi_v) << 16);
        }
        auto r02 = __builtin_amdgcn_permlane32_swap(w[0], w[2], false, false);
        auto r13 = __builtin_amdgcn_permlane32_swap(w[1], w[3], false, false);
        unsigned int frag_words[4] = {(unsigned int)r02[0], (unsigned int)r13[0],
                                      (unsigned int)r02[1], (unsigned int)r13[1]};
        pfrag[s] = *reinterpret_cast<bf16x8*>(frag_words);
      }

      // ---- O^T += V^T P^T : A = V^T (lane holds V[kv-run][own d]) via
      // hardware transpose reads from the panel image --------------------
      {
        const unsigned vt_base = lds_addr(sm->vt);
        const int lam = ln31 & 15;
        const unsigned lane_off = ((lam >> 2) * 16 + 4 * (lam & 3)) * 2;
#pragma unroll
        for (int dblk = 0; dblk < NDBLK; ++dblk) {
          const unsigned pan = (unsigned)((dblk * 2 + (ln31 >> 4))
                                          * (KVBLK * 16)) * 2;
          // small batches (one fragment per wait) keep this kernel under
          // the 256-VGPR 2-waves/SIMD line
#pragma unroll
          for (int s = 0; s < KVBLK / 16; ++s) {
            const unsigned a = pan + (16 * s + 8 * hi) * 32 + lane_off;
            uintx2v r0 = tr_read_b64(vt_base + a);
            uintx2v r1 = tr_read_b64(vt_base + a + 128);
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_sched_barrier(0);
            unsigned w[4] = {r0[0], r0[1], r1[0], r1[1]};
            bf16x8 va = *reinterpret_cast<bf16x8*>(w);
            acc_o[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                va, pfrag[s], acc_o[dblk], 0, 0, 0);
          }
        }
      }
      __syncthreads();
    }

    // ---- epilogue: normalize, write O (scatter by q row) + lse ----------
    const float l_safe = (l_run > 0.f) ? l_run : 1.f;
    const float inv_l = 1.f / l_safe;
    if (qg < Tq) {
      unsigned short* orow = o + q_base + (long)qg * Hq * D;
      // PV output: col = ln31 = this lane's q row; rows crow(r,hi) = d.
      // crow runs in 4-consecutive groups (r&3), so pack 4 bf16 -> one
      // 8-byte store (16 stores/lane instead of 64 scalar).
#pragma unroll
      for (int dblk = 0; dblk < NDBLK; ++dblk) {
#pragma unroll
        for (int r4 = 0; r4 < 4; ++r4) {
          unsigned int w[2];
#pragma unroll
          for (int i = 0; i < 2; ++i) {
            const int r = r4 * 4 + 2 * i;
            w[i] = (unsigned int)f32_to_bf16(acc_o[dblk][r] * inv_l) |
                   ((unsigned int)f32_to_bf16(acc_o[dblk][r + 1] * inv_l) << 16);
          }
          const int d = dblk * 32 + crow(r4 * 4, hi);
          *reinterpret_cast<uint2*>(orow + d) = make_uint2(w[0], w[1]);
        }
      }
      if (hi == 0) {
        lse[(((long)b * Hq) + h) * Tq + qg] =
            (m_run == -INFINITY) ? 0.f : m_run * 0.6931471806f + __logf(l_safe);
      }
    }
  }
};

template <int D, int ABL = 0>
__global__ __launch_bounds__(NWAVES * 64, 2) void attn_fwd_kernel(
    const unsigned short* q, const unsigned short* k, const unsigned short* v,
    unsigned short* o, float* lse, int B, int Tq, int Tkv, int q_off, int Hq,
    int Hkv, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  AttnFwdKernel<D, ABL>::run(q, k, v, o, lse, B, Tq, Tkv, q_off, Hq, Hkv,
                             scale, smem_raw);
}

}  // namespace attn

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool causal,
                                    long q_offset) {
  // q: [B, Tq, Hq, D]; k/v: [B, Tkv, Hkv, D]. q_offset = global key
  // position of q row 0 (context parallelism); Tq == Tkv, q_offset == 0 is
  // the classic square causal case.
  TORCH_CHECK(causal, "attn_fwd: only causal attention is implemented");
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16 && q.dim() == 4
              && q.is_contiguous(), "q must be contiguous bf16 [B,T,Hq,D]");
  TORCH_CHECK(k.is_contiguous() && v.is_contiguous());
  const int B = q.size(0), Tq = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Tkv = k.size(1), Hkv = k.size(2);
  TORCH_CHECK(Hq % Hkv == 0, "GQA requires Hq % Hkv == 0");
  TORCH_CHECK(D == 64 || D == 128, "attn_fwd: head_dim must be 64 or 128");
  TORCH_CHECK(q_offset >= 0 && q_offset + Tq <= Tkv + Tq,
              "q_offset out of range");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, Hq, Tq}, q.options().dtype(torch::kFloat32));
  const float scale = 1.0f / sqrtf((float)D);
  const int n_qblk = (Tq + attn::WG_Q - 1) / attn::WG_Q;
  dim3 grid(n_qblk, Hq, B);
  auto stream = at::cuda::getCurrentHIPStream();
  const size_t smem = sizeof(unsigned short) * (attn::KVBLK * D + D * attn::KVBLK);
  if (D == 128) {
    hipLaunchKernelGGL(attn::attn_fwd_kernel<128>, grid, dim3(attn::NWAVES * 64), smem, stream,
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (unsigned short*)o.data_ptr(), lse.data_ptr<float>(),
                       B, Tq, Tkv, (int)q_offset, Hq, Hkv, scale);
  } else {
    hipLaunchKernelGGL(attn::attn_fwd_kernel<64>, grid, dim3(attn::NWAVES * 64), smem, stream,
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (unsigned short*)o.data_ptr(), lse.data_ptr<float>(),
                       B, Tq, Tkv, (int)q_offset, Hq, Hkv, scale);
  }
  HIP_CHECK_KERNEL();
  return {o, lse};
}

// Ablation entry (D=128 only): mode 0 full / 1 no-Vt-staging / 2 no-staging
// / 3 no-softmax. Modes>0 produce wrong outputs; timing only.
std::vector<torch::Tensor> attn_fwd_ablate(torch::Tensor q, torch::Tensor k,
                                           torch::Tensor v, long mode) {
  const int B = q.size(0), T = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hkv = k.size(2);
  TORCH_CHECK(D == 128);
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, Hq, T}, q.options().dtype(torch::kFloat32));
  const float scale = 1.0f / sqrtf((float)D);
  dim3 grid((T + attn::WG_Q - 1) / attn::WG_Q, Hq, B);
  auto stream = at::cuda::getCurrentHIPStream();
  const size_t smem = sizeof(unsigned short) * (attn::KVBLK * D + D * attn::KVBLK);
  auto launch = [&](auto kfn) {
    hipLaunchKernelGGL(kfn, grid, dim3(attn::NWAVES * 64), smem, stream,
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (unsigned short*)o.data_ptr(), lse.data_ptr<float>(),
                       B, T, T, 0, Hq, Hkv, scale);
  };
  switch (mode) {
    case 0: launch(attn::attn_fwd_kernel<128, 0>); break;
    case 1: launch(attn::attn_fwd_kernel<128, 1>); break;
    case 2: launch(attn::attn_fwd_kernel<128, 2>); break;
    case 3: launch(attn::attn_fwd_kernel<128, 3>); break;
    default: TORCH_CHECK(false, "mode 0-3");
  }
  HIP_CHECK_KERNEL();
  return {o, lse};
}

// ---------------------------------------------------------------------------
// MFMA layout probes: empirical verification of the fragment maps used above
// (guide §3: always check with ASYMMETRIC operands).
// a: [32,16] f32, b: [16,32] f32 -> d: [32,32] = a @ b
__global__ void mfma_probe_32x32x16_kernel(const float* a, const float* b,
                                           float* d) {
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5, ln31 = lane & 31;
  bf16x8 af, bf;
#pragma unroll
  for (int m = 0; m < 8; ++m) {
    af[m] = (__bf16)a[ln31 * 16 + hi * 8 + m];       // A[row=ln31][k=hi*8+m]
    bf[m] = (__bf16)b[(hi * 8 + m) * 32 + ln31];     // B[k][col=ln31]
  }
  floatx16 acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    d[row * 32 + ln31] = acc[r];
  }
}

// ---------------------------------------------------------------------------
// ds_read_b64_tr_b16 semantics probe: stage bf16 values v[i] = i for
// i in [0,512) into LDS, have each lane of one wave issue the transpose read
// with a configurable per-lane address, dump the 4 returned elements per
// lane. mode selects the addressing convention under test:
//   0: addr = lane*8                    (contiguous 8B rows per lane)
//   1: addr = (lane&15)*2 + (lane>>4)*128   (strided column interpretation)
//   2: addr = 0                         (uniform)
typedef __attribute__((ext_vector_type(2))) unsigned int uintx2;

__global__ void tr16_probe_kernel(unsigned short* out, int mode) {
  __shared__ unsigned short lds[512];
  const int tid = threadIdx.x;
  for (int i = tid; i < 512; i += blockDim.x) lds[i] = (unsigned short)i;
  __syncthreads();
  if (tid >= 64) return;
  unsigned base = (unsigned)(unsigned long long)(void*)&lds[0];
  unsigned addr;
  if (mode == 0) addr = base + tid * 8;
  else if (mode == 1) addr = base + (tid & 15) * 2 + (tid >> 4) * 128;
  else addr = base;
  uintx2 r;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(r) : "v"(addr) : "memory");
  const unsigned short* rr = reinterpret_cast<const unsigned short*>(&r);
  for (int j = 0; j < 4; ++j) out[tid * 4 + j] = rr[j];
}

torch::Tensor tr16_probe(long mode) {
  auto out = torch::zeros({64, 4}, torch::dtype(torch::kInt16).device(torch::kCUDA));
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (unsigned short*)out.data_ptr(), (int)mode);
  HIP_CHECK_KERNEL();
  return out;
}

torch::Tensor mfma_probe_32x32x16(torch::Tensor a, torch::Tensor b) {
  auto d = torch::zeros({32, 32}, a.options());
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_probe_32x32x16_kernel, dim3(1), dim3(64), 0, stream,
                     a.data_ptr<float>(), b.data_ptr<float>(),
                     d.data_ptr<float>());
  HIP_CHECK_KERNEL();
  return d;
}
