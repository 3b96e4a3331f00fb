// K1 v2: flash attention fwd + bwd for CDNA4/gfx950, GQA-aware.
//
// Same math and fragment maps as attention_fwd.hip / attention_bwd.hip
// (v1), restructured for occupancy and stall behavior after r1 profiling
// showed both backward kernels allocate the whole 512-register file
// (1 wave/SIMD: dq 256v+136a, dkdv 256v+256a) and the forward stages
// synchronously (HBM latency exposed every tile):
//
//   - 8-wave (512-thread) workgroups everywhere: 2 waves/SIMD co-hide
//     LDS/HBM latency (MI355X_MICROARCH "Two waves per SIMD"); register
//     budget forced <= 256 via __launch_bounds__(512, 2).
//   - T14 issue-early/write-late staging with ONE register set and ONE
//     LDS buffer in all three kernels (HBM latency hides under MFMAs).
//   - ds_read_b64_tr_b16 batches of 8 with a single lgkmcnt drain per
//     d-block (v1 fwd drained per MFMA pair).
//   - heavy-first block order for the causal triangle (fwd/dq blocks with
//     the most kv tiles dispatch first).
//   - per-wave fully-masked-tile skip (waves outside the causal band for
//     a tile skip its MFMAs but keep the barrier schedule).
//   - head_dim 80 (the reference 2.7B shape, gpt2_model.py:446-461)
//     supported natively: contraction runs D/16 = 5 k-steps; LDS images
//     keep the 128-column layout (stride DS) with the d >= 80 panels
//     zeroed once; output stores masked to d < 80.
//
// Reference parity: replaces flash-attn 2.8.3 / SDPA
// (src/modalities/models/gpt2/gpt2_model.py:595-658).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPCachingAllocator.h>
#include <c10/hip/HIPStream.h>

#include "attn_common.h"

namespace attn2 {

using attnc::c_layout_to_frags;
using attnc::crow;
using attnc::lds_addr;
using attnc::swz;
using attnc::tr_read_b64;

constexpr int KVBLK = 64;   // fwd/dq kv tile rows
constexpr int QBLK = 32;    // q rows per wave (fwd/dq) / per LDS tile (dkdv)
constexpr int NW = 8;       // waves per workgroup
constexpr int NT = NW * 64; // threads per workgroup

template <int D> struct Geom {
  static constexpr int DS = (D == 64) ? 64 : 128;  // logical column count
  // Row pitch padded by one b128 access width (guide G4): row-major reads
  // at (row = lane, col fixed) land on distinct banks with AFFINE
  // addresses — the compiler folds the whole k-loop into ds offset
  // immediates (the XOR swizzle costs ~5 address VALU per read).
  static constexpr int DP = DS + 8;                // padded LDS row pitch
  static constexpr int NDSTEP = D / 16;            // contraction k-steps
  static constexpr int NDBLK = (D + 31) / 32;      // output 32-col blocks
  static_assert(D % 16 == 0 && D <= 128, "head_dim must be mult of 16, <=128");
};

// ---------------------------------------------------------------------------
// Forward: 8 waves x 32 q rows = 256 q rows per WG; KV tiles of 64 in LDS
// (K row-major XOR-swizzled; V as [DS/16][64][16] panels for tr reads).
// Swapped QK^T (S^T = K Q^T) keeps softmax lane-local (guide T12);
// defer-max rescale skip (T13, THR=8); exp2-domain state.
template <int D_>
struct FwdV2 {
  using G = Geom<D_>;
  static constexpr int D = D_, DS = G::DS, DP = G::DP;
  static constexpr int NDSTEP = G::NDSTEP, NDBLK = G::NDBLK;
  static constexpr int WG_Q = QBLK * NW;

  // NSUB kv sub-tiles of KVBLK rows staged per barrier pair (amortizes
  // the two per-stage __syncthreads + load issue over 2x the MFMA work)
  static constexpr int NSUB = 2;
  static constexpr int TKV = KVBLK * NSUB;
  static constexpr int PANEL = (DS / 16) * KVBLK * 16;  // vt elems per sub
  struct Smem {
    unsigned short k[NSUB * KVBLK * DP];
    unsigned short vt[NSUB * PANEL];
  };
  // double-buffered: single-barrier T14 loop (see DKDVV2)
  static constexpr int BUF_BYTES = (int)((sizeof(Smem) + 15) / 16 * 16);
  static constexpr size_t SMEM_BYTES = (size_t)BUF_BYTES * 2;
  static constexpr int NCHS = TKV * D / 8;         // 16B chunks per tensor
  static constexpr int NCH = (NCHS + NT - 1) / NT; // chunks per thread

  struct Stage {
    shortx8 kk[NCH], vv[NCH];
  };

  // vp = V row pitch (elements between consecutive (b,t) rows): Hkv*D for
  // a standalone tensor, Cq+2*Ckv when V is read in place from the joint
  // QKV activation (kills the per-layer V-slice copy).
  static __device__ void stage_load(Stage& r, const unsigned short* k,
                                    const unsigned short* v, long kv_base,
                                    long v_base, long vp,
                                    int kv0, int Tkv, int Hkv) {
    const int tid = threadIdx.x;
#pragma unroll
    for (int it = 0; it < NCH; ++it) {
      const int i = tid + it * NT;
      if (NCHS % NT != 0 && i >= NCHS) continue;
      const int r_ = (i * 8) / D, c = (i * 8) % D;
      if (kv0 + r_ < Tkv) {
        r.kk[it] = *reinterpret_cast<const shortx8*>(
            k + kv_base + (long)(kv0 + r_) * Hkv * D + c);
        r.vv[it] = *reinterpret_cast<const shortx8*>(
            v + v_base + (long)(kv0 + r_) * vp + c);
      } else {
#pragma unroll
        for (int m = 0; m < 8; ++m) { r.kk[it][m] = 0; r.vv[it][m] = 0; }
      }
    }
  }

  static __device__ void stage_write(const Stage& r, Smem* sm) {
    const int tid = threadIdx.x;
#pragma unroll
    for (int it = 0; it < NCH; ++it) {
      const int i = tid + it * NT;
      if (NCHS % NT != 0 && i >= NCHS) continue;
      const int r_ = (i * 8) / D, c = (i * 8) % D;
      const int sub = r_ / KVBLK, rin = r_ % KVBLK;
      *reinterpret_cast<shortx8*>(
          &sm->k[sub * (KVBLK * DP) + rin * DP + c]) = r.kk[it];
      *reinterpret_cast<shortx8*>(
          &sm->vt[sub * PANEL + (c >> 4) * (KVBLK * 16) + rin * 16
                  + (c & 15)]) = r.vv[it];
    }
  }

  static __device__ void run(const unsigned short* __restrict__ q,
                             const unsigned short* __restrict__ k,
                             const unsigned short* __restrict__ v, long vp,
                             unsigned short* __restrict__ o,
                             float* __restrict__ lse, int B, int Tq, int Tkv,
                             int q_off, int Hq, int Hkv, float scale,
                             char* smem_raw) {
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int hi = lane >> 5, ln31 = lane & 31;

    // heavy-first: high q blocks have the most kv tiles under causal
    const int qi = (int)gridDim.x - 1 - (int)blockIdx.x;
    const int qblk0 = qi * WG_Q;
    const int h = blockIdx.y, b = blockIdx.z;
    const int hkv = h / (Hq / Hkv);

    const long q_base = (((long)b * Tq) * Hq + h) * D;
    const long kv_base = (((long)b * Tkv) * Hkv + hkv) * D;
    const long v_base = ((long)b * Tkv) * vp + (long)hkv * D;
    const int qg = qblk0 + wid * QBLK + ln31;  // this lane's LOCAL q row
    const int qgl = qg + q_off;                // global vs keys
    const float scale2 = scale * 1.44269504f;  // exp2 fold

    // zero the d >= D panels of the V image once (D=80: panel 5 feeds the
    // dblk-2 MFMAs and must read as 0; panels 6,7 cleared too, cheap)
    if constexpr (D != DS) {
#pragma unroll
      for (int bufi = 0; bufi < 2; ++bufi) {
        Smem* sz = reinterpret_cast<Smem*>(smem_raw + bufi * BUF_BYTES);
#pragma unroll
        for (int sub = 0; sub < NSUB; ++sub)
          for (int i = threadIdx.x; i < (DS / 16 - D / 16) * KVBLK * 16;
               i += NT)
            sz->vt[sub * PANEL + (D / 16) * KVBLK * 16 + i] = 0;
      }
      // first stage_write below is followed by __syncthreads
    }

    bf16x8 qfrag[NDSTEP];
    {
      const unsigned short* qr = q + q_base + (long)qg * Hq * D;
#pragma unroll
      for (int s = 0; s < NDSTEP; ++s) {
        if (qg < Tq) {
          qfrag[s] = *reinterpret_cast<const bf16x8*>(qr + hi * 8 + 16 * s);
        } else {
#pragma unroll
          for (int m = 0; m < 8; ++m) qfrag[s][m] = (__bf16)0.f;
        }
      }
    }

    float m_run = -INFINITY, l_run = 0.f;
    floatx16 acc_o[NDBLK];
#pragma unroll
    for (int dblk = 0; dblk < NDBLK; ++dblk)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc_o[dblk][r] = 0.f;

    const int q_hi_wg = min(qblk0 + WG_Q - 1, Tq - 1) + q_off;
    const int n_tiles = (min(q_hi_wg, Tkv - 1)) / TKV + 1;
    // this wave's own causal bound (skip fully-masked tiles' compute)
    const int q_hi_wave = min(qblk0 + wid * QBLK + QBLK - 1, Tq - 1) + q_off;
    const int q_lo_wave = qblk0 + wid * QBLK + q_off;

    Stage st;
    stage_load(st, k, v, kv_base, v_base, vp, 0, Tkv, Hkv);
    stage_write(st, reinterpret_cast<Smem*>(smem_raw));
    __syncthreads();
    if (n_tiles > 1) stage_load(st, k, v, kv_base, v_base, vp, TKV, Tkv, Hkv);

    const int lam = ln31 & 15;
    const unsigned trb0 = lds_addr(
        reinterpret_cast<Smem*>(smem_raw)->vt)
        + (unsigned)(ln31 >> 4) * (KVBLK * 16 * 2) + (unsigned)hi * 256
        + ((lam >> 2) * 16 + 4 * (lam & 3)) * 2;


    for (int tile = 0; tile < n_tiles; ++tile) {
      const int kv0t = tile * TKV;
      const bool stage_active = (kv0t <= q_hi_wave);
      const unsigned boff = (unsigned)(tile & 1) * BUF_BYTES;
      Smem* const sm = reinterpret_cast<Smem*>(smem_raw + boff);
      const unsigned trb = trb0 + boff;
      // single-barrier T14: stage tile+1 into the other buffer and issue
      // tile+2's loads BEFORE this tile's compute
      if (tile + 1 < n_tiles) {
        stage_write(st, reinterpret_cast<Smem*>(
            smem_raw + (unsigned)(~tile & 1) * BUF_BYTES));
        if (tile + 2 < n_tiles)
          stage_load(st, k, v, kv_base, v_base, vp, (tile + 2) * TKV,
                     Tkv, Hkv);
      }

      if (stage_active) attnc::static_for<NSUB>([&](auto sub_) {
        constexpr int sub = decltype(sub_)::value;
        const int kv0 = kv0t + KVBLK * sub;
        if (kv0 > q_hi_wave) return;  // sub fully masked for this wave
        // ---- S^T = K Q^T ------------------------------------------------
        floatx16 s0, s1;
#pragma unroll
        for (int r = 0; r < 16; ++r) { s0[r] = 0.f; s1[r] = 0.f; }
#pragma unroll
        for (int s = 0; s < NDSTEP; ++s) {
          const int col = hi * 8 + 16 * s;
          const int r0 = ln31, r1 = ln31 + 32;
          bf16x8 ka = *reinterpret_cast<const bf16x8*>(
              &sm->k[sub * (KVBLK * DP) + r0 * DP + col]);
          bf16x8 kb = *reinterpret_cast<const bf16x8*>(
              &sm->k[sub * (KVBLK * DP) + r1 * DP + col]);
          s0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[s], s0, 0, 0, 0);
          s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kb, qfrag[s], s1, 0, 0, 0);
        }

        // ---- scale + causal mask; per-lane P rows (q = ln31).
        // Interior fast path: row max taken on RAW scores (one scale
        // multiply at the end instead of 32), P computed later as
        // exp2(fma(s, scale2, -m)) — saves ~64 VALU/tile.
        float p[32];
        float tmax = -INFINITY;
        bool raw_scores = false;
        if (kv0 + KVBLK - 1 <= q_lo_wave && kv0 + KVBLK <= Tkv && qg < Tq) {
          raw_scores = true;
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            p[r] = s0[r];
            p[16 + r] = s1[r];
            tmax = fmaxf(tmax, fmaxf(s0[r], s1[r]));
          }
          tmax *= scale2;
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
        tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));

        // ---- online softmax with defer-max (T13) ------------------------
        constexpr float RESCALE_THR = 8.0f;
        const bool defer = __all(tmax - m_run <= RESCALE_THR);
        if (!defer) {
          const float m_new = fmaxf(m_run, tmax);
          const float m_safe = (m_new == -INFINITY) ? 0.f : m_new;
          const float alpha = (m_run == -INFINITY)
              ? 0.f : __builtin_amdgcn_exp2f(m_run - m_safe);
          l_run *= alpha;
#pragma unroll
          for (int dblk = 0; dblk < NDBLK; ++dblk)
#pragma unroll
            for (int r = 0; r < 16; ++r) acc_o[dblk][r] *= alpha;
          m_run = m_new;
        }
        const float m_safe2 = (m_run == -INFINITY) ? 0.f : m_run;
        float psum = 0.f;
        if (raw_scores) {
#pragma unroll
          for (int r = 0; r < 32; ++r) {
            p[r] = __builtin_amdgcn_exp2f(fmaf(p[r], scale2, -m_safe2));
            psum += p[r];
          }
        } else {
#pragma unroll
          for (int r = 0; r < 32; ++r) {
            p[r] = __builtin_amdgcn_exp2f(p[r] - m_safe2);
            psum += p[r];
          }
        }
        psum += __shfl_xor(psum, 32, 64);
        l_run += psum;

        // ---- P -> bf16 PV fragments (T12) -------------------------------
        bf16x8 pfrag[KVBLK / 16];
        c_layout_to_frags(p, pfrag, KVBLK / 16);

        // ---- O^T += V^T P^T: batched tr reads, one drain per d-block;
        // one tile-invariant base VGPR + ds offset immediates ----
        attnc::static_for<NDBLK>([&](auto dblk_) {
          constexpr int dblk = decltype(dblk_)::value;
          uintx2 rv[2 * (KVBLK / 16)];
          attnc::static_for<KVBLK / 16>([&](auto s_) {
            constexpr int s = decltype(s_)::value;
            constexpr unsigned a = sub * (PANEL * 2)
                + dblk * 2 * (KVBLK * 16) * 2 + s * 512;
            rv[2 * s] = attnc::tr_read_b64_off<a>(trb);
            rv[2 * s + 1] = attnc::tr_read_b64_off<a + 128>(trb);
          });
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
#pragma unroll
          for (int s = 0; s < KVBLK / 16; ++s) {
            unsigned w[4] = {rv[2 * s][0], rv[2 * s][1],
                             rv[2 * s + 1][0], rv[2 * s + 1][1]};
            bf16x8 va = *reinterpret_cast<bf16x8*>(w);
            acc_o[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                va, pfrag[s], acc_o[dblk], 0, 0, 0);
          }
        });
      });  // sub / stage_active

      __syncthreads();  // tile reads done AND tile+1 writes landed
    }

    // ---- epilogue -------------------------------------------------------
    const float l_safe = (l_run > 0.f) ? l_run : 1.f;
    const float inv_l = 1.f / l_safe;
    if (qg < Tq) {
      unsigned short* orow = o + q_base + (long)qg * Hq * D;
#pragma unroll
      for (int dblk = 0; dblk < NDBLK; ++dblk) {
#pragma unroll
        for (int r4 = 0; r4 < 4; ++r4) {
          const int d = dblk * 32 + crow(r4 * 4, hi);
          if (D != DS && d >= D) continue;  // head_dim 80 tail mask
          unsigned int w[2];
#pragma unroll
          for (int i = 0; i < 2; ++i) {
            const int r = r4 * 4 + 2 * i;
            w[i] = attnc::cvt_pk_bf16(acc_o[dblk][r] * inv_l,
                                      acc_o[dblk][r + 1] * inv_l);
          }
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

// ---------------------------------------------------------------------------
// dQ: 8 waves x 32 q rows = 256 q rows per WG; kv tiles of 64 in LDS
// (K row + V row as S^T/dP^T A-operands; Kt panels as the dQ^T A-operand).
// The kv tile is processed in two 32-row halves so only one (s, dp)
// accumulator pair is live at a time (register budget).
template <int D_>
struct DQV2 {
  using G = Geom<D_>;
  static constexpr int D = D_, DS = G::DS, DP = G::DP;
  static constexpr int NDSTEP = G::NDSTEP, NDBLK = G::NDBLK;
  static constexpr int WG_Q = QBLK * NW;

  // NSUB kv sub-tiles of KVBLK rows staged per barrier pair (see DKDVV2).
  // K lives ONLY as the [D/16][KVBLK][16] panel image: the S^T A-operand
  // reads 8 consecutive d of one kv row, which is contiguous within a
  // panel (<=2-way bank conflict) — dropping the row-major K image makes
  // room for double buffering (single-barrier T14 loop).
  static constexpr int NSUB = 2;
  static constexpr int TKV = KVBLK * NSUB;
  static constexpr int PANEL = (DS / 16) * KVBLK * 16;  // kt elems per sub
  struct Smem {
    unsigned short v[NSUB * KVBLK * DP];
    unsigned short kt[NSUB * PANEL];
  };
  static constexpr int BUF_BYTES = (int)((sizeof(Smem) + 15) / 16 * 16);
  static constexpr size_t SMEM_BYTES = (size_t)BUF_BYTES * 2;
  static constexpr int NCHS = TKV * D / 8;
  static constexpr int NCH = (NCHS + NT - 1) / NT;

  struct Stage {
    shortx8 kk[NCH], vv[NCH];
  };

  static __device__ void stage_load(Stage& r, const unsigned short* k,
                                    const unsigned short* v, long kv_base,
                                    long v_base, long vp,
                                    int kv0, int Tkv, int Hkv) {
    const int tid = threadIdx.x;
#pragma unroll
    for (int it = 0; it < NCH; ++it) {
      const int i = tid + it * NT;
      if (NCHS % NT != 0 && i >= NCHS) continue;
      const int r_ = (i * 8) / D, c = (i * 8) % D;
      if (kv0 + r_ < Tkv) {
        r.kk[it] = *reinterpret_cast<const shortx8*>(
            k + kv_base + (long)(kv0 + r_) * Hkv * D + c);
        r.vv[it] = *reinterpret_cast<const shortx8*>(
            v + v_base + (long)(kv0 + r_) * vp + c);
      } else {
#pragma unroll
        for (int m = 0; m < 8; ++m) { r.kk[it][m] = 0; r.vv[it][m] = 0; }
      }
    }
  }

  static __device__ void stage_write(const Stage& r, Smem* sm) {
    const int tid = threadIdx.x;
#pragma unroll
    for (int it = 0; it < NCH; ++it) {
      const int i = tid + it * NT;
      if (NCHS % NT != 0 && i >= NCHS) continue;
      const int r_ = (i * 8) / D, c = (i * 8) % D;
      const int sub = r_ / KVBLK, rin = r_ % KVBLK;
      *reinterpret_cast<shortx8*>(
          &sm->v[sub * (KVBLK * DP) + rin * DP + c]) = r.vv[it];
      *reinterpret_cast<shortx8*>(
          &sm->kt[sub * PANEL + (c >> 4) * (KVBLK * 16) + rin * 16
                  + (c & 15)]) = r.kk[it];
    }
  }

  static __device__ void run(const unsigned short* q, const unsigned short* k,
                             const unsigned short* v, long vp,
                             const unsigned short* dout,
                             const float* lse, const float* delta,
                             unsigned short* dq, int B, int Tq, int Tkv,
                             int q_off, int Hq, int Hkv, float scale,
                             char* smem_raw) {
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int hi = lane >> 5, ln31 = lane & 31;
    const int h = blockIdx.y, b = blockIdx.z;
    const int hkv = h / (Hq / Hkv);
    const int qi = (int)gridDim.x - 1 - (int)blockIdx.x;  // heavy-first
    const int qblk0 = qi * WG_Q;
    const int qg = qblk0 + wid * QBLK + ln31;  // LOCAL q row
    const int qgl = qg + q_off;

    const long q_base = (((long)b * Tq) * Hq + h) * D;
    const long kv_base = (((long)b * Tkv) * Hkv + hkv) * D;
    const long v_base = ((long)b * Tkv) * vp + (long)hkv * D;

    if constexpr (D != DS) {
#pragma unroll
      for (int bufi = 0; bufi < 2; ++bufi) {
        Smem* sz = reinterpret_cast<Smem*>(smem_raw + bufi * BUF_BYTES);
#pragma unroll
        for (int sub = 0; sub < NSUB; ++sub)
          for (int i = threadIdx.x; i < (DS / 16 - D / 16) * KVBLK * 16;
               i += NT)
            sz->kt[sub * PANEL + (D / 16) * KVBLK * 16 + i] = 0;
      }
    }

    bf16x8 qfrag[NDSTEP], dofrag[NDSTEP];
    {
      const unsigned short* qr = q + q_base + (long)qg * Hq * D;
      const unsigned short* dor = dout + q_base + (long)qg * Hq * D;
#pragma unroll
      for (int s = 0; s < NDSTEP; ++s) {
        if (qg < Tq) {
          qfrag[s] = *reinterpret_cast<const bf16x8*>(qr + hi * 8 + 16 * s);
          dofrag[s] = *reinterpret_cast<const bf16x8*>(dor + hi * 8 + 16 * s);
        } else {
#pragma unroll
          for (int m = 0; m < 8; ++m) { qfrag[s][m] = (__bf16)0.f;
                                        dofrag[s][m] = (__bf16)0.f; }
        }
      }
    }
    const float my_lse2 = ((qg < Tq) ? lse[(((long)b * Hq) + h) * Tq + qg]
                                      : 0.f) * 1.44269504f;
    const float scale2 = scale * 1.44269504f;
    const float my_delta = (qg < Tq)
        ? delta[(((long)b * Hq) + h) * Tq + qg] : 0.f;

    floatx16 acc_dq[NDBLK];
#pragma unroll
    for (int dblk = 0; dblk < NDBLK; ++dblk)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc_dq[dblk][r] = 0.f;

    const int q_hi_wg = min(qblk0 + WG_Q - 1, Tq - 1) + q_off;
    const int n_tiles = min(q_hi_wg, Tkv - 1) / TKV + 1;
    const int q_hi_wave = min(qblk0 + wid * QBLK + QBLK - 1, Tq - 1) + q_off;
    const int q_lo_wave = qblk0 + wid * QBLK + q_off;

    Stage st;
    stage_load(st, k, v, kv_base, v_base, vp, 0, Tkv, Hkv);
    stage_write(st, reinterpret_cast<Smem*>(smem_raw));
    __syncthreads();
    if (n_tiles > 1) stage_load(st, k, v, kv_base, v_base, vp, TKV, Tkv, Hkv);

    const int lam = ln31 & 15;
    const unsigned trb0 = lds_addr(reinterpret_cast<Smem*>(smem_raw)->kt)
        + (unsigned)(ln31 >> 4) * (KVBLK * 16 * 2) + (unsigned)hi * 256
        + ((lam >> 2) * 16 + 4 * (lam & 3)) * 2;


    for (int tile = 0; tile < n_tiles; ++tile) {
      const int kv0t = tile * TKV;
      const bool stage_active = (kv0t <= q_hi_wave);
      const unsigned boff = (unsigned)(tile & 1) * BUF_BYTES;
      Smem* const sm = reinterpret_cast<Smem*>(smem_raw + boff);
      const unsigned trb = trb0 + boff;
      // single-barrier T14 (see FwdV2)
      if (tile + 1 < n_tiles) {
        stage_write(st, reinterpret_cast<Smem*>(
            smem_raw + (unsigned)(~tile & 1) * BUF_BYTES));
        if (tile + 2 < n_tiles)
          stage_load(st, k, v, kv_base, v_base, vp, (tile + 2) * TKV,
                     Tkv, Hkv);
      }

      if (stage_active) attnc::static_for<NSUB>([&](auto sub_) {
        constexpr int sub = decltype(sub_)::value;
        const int kv0 = kv0t + KVBLK * sub;
        if (kv0 > q_hi_wave) return;  // sub fully masked for this wave
        attnc::static_for<2>([&](auto half_) {
          constexpr int half = decltype(half_)::value;
          const int kv0h = kv0 + 32 * half;
          // S^T = K Q^T ; dP^T = V dO^T  (col = q = ln31, rows = kv half)
          floatx16 s_h, dp_h;
#pragma unroll
          for (int r = 0; r < 16; ++r) { s_h[r] = 0.f; dp_h[r] = 0.f; }
#pragma unroll
          for (int s = 0; s < NDSTEP; ++s) {
            const int col = hi * 8 + 16 * s;
            const int r_ = 32 * half + ln31;
            // K row-read from the panel image: 8 consecutive d of row r_
            // are contiguous within panel s (<=2-way bank conflict)
            bf16x8 ka = *reinterpret_cast<const bf16x8*>(
                &sm->kt[sub * PANEL + s * (KVBLK * 16) + r_ * 16 + hi * 8]);
            bf16x8 va = *reinterpret_cast<const bf16x8*>(
                &sm->v[sub * (KVBLK * DP) + r_ * DP + col]);
            s_h = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[s], s_h,
                                                          0, 0, 0);
            dp_h = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, dofrag[s], dp_h,
                                                           0, 0, 0);
          }

          // dS'^T = scale * P .* (dP - delta), P = exp2(scale2*S' - lse2);
          // written back into dp_h (register-frugal)
          if (kv0h + 31 <= q_lo_wave && kv0h + 32 <= Tkv && qg < Tq) {
#pragma unroll
            for (int r = 0; r < 16; ++r) {
              const float p = __builtin_amdgcn_exp2f(
                  fmaf(s_h[r], scale2, -my_lse2));
              dp_h[r] = scale * p * (dp_h[r] - my_delta);
            }
          } else {
#pragma unroll
            for (int r = 0; r < 16; ++r) {
              const int kg = kv0h + crow(r, hi);
              const float p = (kg <= qgl && kg < Tkv && qg < Tq)
                  ? __builtin_amdgcn_exp2f(fmaf(s_h[r], scale2, -my_lse2))
                  : 0.f;
              dp_h[r] = scale * p * (dp_h[r] - my_delta);
            }
          }
          bf16x8 dsfrag[2];
          attnc::c16_to_frags(dp_h, dsfrag);

          // dQ^T += Kt dS'^T (contraction over this kv half: k-steps
          // 2*half, 2*half+1 of the Kt panel image; base VGPR + offset
          // immediates)
          attnc::static_for<NDBLK>([&](auto dblk_) {
            constexpr int dblk = decltype(dblk_)::value;
            uintx2 rk[4];
            attnc::static_for<2>([&](auto s2_) {
              constexpr int s2 = decltype(s2_)::value;
              constexpr unsigned a = sub * (PANEL * 2)
                  + dblk * 2 * (KVBLK * 16) * 2 + (2 * half + s2) * 512;
              rk[2 * s2] = attnc::tr_read_b64_off<a>(trb);
              rk[2 * s2 + 1] = attnc::tr_read_b64_off<a + 128>(trb);
            });
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_sched_barrier(0);
#pragma unroll
            for (int s2 = 0; s2 < 2; ++s2) {
              unsigned w[4] = {rk[2 * s2][0], rk[2 * s2][1],
                               rk[2 * s2 + 1][0], rk[2 * s2 + 1][1]};
              bf16x8 kta = *reinterpret_cast<bf16x8*>(w);
              acc_dq[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                  kta, dsfrag[s2], acc_dq[dblk], 0, 0, 0);
            }
          });
        });  // half
      });  // sub / stage_active

      __syncthreads();  // tile reads done AND tile+1 writes landed
    }

    if (qg < Tq) {
      unsigned short* dqr = dq + q_base + (long)qg * Hq * D;
#pragma unroll
      for (int dblk = 0; dblk < NDBLK; ++dblk)
#pragma unroll
        for (int r4 = 0; r4 < 4; ++r4) {
          const int d = dblk * 32 + crow(r4 * 4, hi);
          if (D != DS && d >= D) continue;
          unsigned int w[2];
#pragma unroll
          for (int i = 0; i < 2; ++i) {
            const int r = r4 * 4 + 2 * i;
            w[i] = attnc::cvt_pk_bf16(acc_dq[dblk][r], acc_dq[dblk][r + 1]);
          }
          *reinterpret_cast<uint2*>(dqr + d) = make_uint2(w[0], w[1]);
        }
    }
  }
};

// ---------------------------------------------------------------------------
// dKdV: MODE-parametrized.
//   MODE 0 (fused, D=64/80): 8 waves x 32 kv rows = 256 kv rows per WG;
//     computes dK and dV in one pass (fits the 256-register budget at
//     D<=80).
//   MODE 1 (dV-only) / MODE 2 (dK-only), D=128: 4 waves x 32 kv rows =
//     128 kv rows per WG. The fused D=128 kernel needs ~290 registers
//     (acc 128 + K/V fragments 64 + softmax/frags/staging) and spilled
//     ~116 VGPRs at 1 wave/SIMD-equivalent cost; the split recomputes S
//     (+25% MFMA) but each kernel is register-clean at 2-3 waves/SIMD and
//     both run concurrently on separate streams.
// K,V live in REGISTERS (lane owns kv row wave_kv0 + ln31); q tiles of 32
// in LDS, single buffer, T14 issue-early/write-late. Iteration space =
// (GQA q-head) x (q tiles).
//   S [q][kv]  = mfma(A=Q_lds,  B=K_regs)    col = kv = ln31, rows q = crow
//   dP[q][kv]  = mfma(A=dO_lds, B=V_regs)
//   dV[kv][d] += mfma(A=frags(P),   B=dOt)   col = d = ln31, rows kv = crow
//   dK[kv][d] += mfma(A=frags(dS'), B=Qt)
template <int D_, int MODE>
struct DKDVV2 {
  using G = Geom<D_>;
  static constexpr int D = D_, DS = G::DS;
  static constexpr int NDSTEP = G::NDSTEP, NDBLK = G::NDBLK;
  static constexpr bool DO_DV = (MODE != 2), DO_DK = (MODE != 1);
  static constexpr int NWV = (MODE == 0) ? 8 : 4;   // waves per WG
  static constexpr int NTV = NWV * 64;
  static constexpr int WG_KV = QBLK * NWV;

  // Two 32-row q sub-tiles are staged per barrier pair (NSUB): amortizes
  // the per-stage __syncthreads pair + global-load issue over 2x the MFMA
  // work (the per-tile fixed costs dominated at 32-row tiles). The fused
  // D=80 kernel sits exactly at the 256-register line — doubling its
  // staging registers would spill into the hot loop, so it stays at 1.
  // The fused D=80 kernel sits exactly at the 256-register line: the
  // NSUB staging registers spill into the hot loop (184 B/lane measured,
  // even with halved tr batches + compressed cursors), so it stays at
  // one sub-tile. The A/B against the register-clean split dV/dK pair at
  // NSUB=2 still favors the fused kernel (1.95 vs 2.01 ms/layer,
  // profiles/r2_bench_2p7b_1gpu.md).
  static constexpr int NSUB = (MODE == 0 && D_ == 80) ? 1 : 2;
  static constexpr bool TRHALF = false;
  static constexpr int TQ = QBLK * NSUB;               // q rows per stage
  // manual LDS carve (only the images this mode reads); each image is
  // NSUB consecutive per-sub blocks
  static constexpr int DP = G::DP;                     // padded row pitch
  static constexpr int PANEL = (DS / 16) * QBLK * 16;  // elements per sub
  static constexpr int O_QROW = 0;
  static constexpr int O_DOROW = O_QROW + NSUB * QBLK * DP;
  static constexpr int O_QT = O_DOROW + (DO_DK ? NSUB * QBLK * DP : 0);
  static constexpr int O_DOT = O_QT + (DO_DK ? NSUB * PANEL : 0);
  static constexpr int O_END = O_DOT + (DO_DV ? NSUB * PANEL : 0);
  static constexpr int O_STATS = (O_END * 2 + 15) / 16 * 16;  // bytes, 16-al
  // Double-buffered stages run the canonical single-barrier T14 loop
  // (write tile t+1 into the OTHER buffer + issue t+2 loads BEFORE the
  // compute of t; one __syncthreads per stage). The dK-only kernel runs
  // 2 workgroups/CU and cannot fit two buffers in its LDS share.
  static constexpr bool DBUF = (MODE != 2);
  static constexpr int BUF_BYTES =
      (O_STATS + (DO_DK ? 2 : 1) * NSUB * 32 * 4 + 15) / 16 * 16;
  static constexpr size_t SMEM_BYTES = (size_t)BUF_BYTES * (DBUF ? 2 : 1);

  static constexpr int NCHS = TQ * D / 8;
  static constexpr int NCH = (NCHS + NTV - 1) / NTV;

  struct Stage {
    shortx8 qq[NCH], dd[NCH];
    float stat;
  };

  static __device__ void stage_load(Stage& r, const unsigned short* q,
                                    const unsigned short* dout, long q_base,
                                    const float* lse_h, const float* delta_h,
                                    int q0, int Tq, int Hq) {
    const int tid = threadIdx.x;
#pragma unroll
    for (int it = 0; it < NCH; ++it) {
      const int i = tid + it * NTV;
      if (NCHS % NTV != 0 && i >= NCHS) continue;
      const int row = (i * 8) / D, c = (i * 8) % D;
      if (q0 + row < Tq) {
        r.qq[it] = *reinterpret_cast<const shortx8*>(
            q + q_base + (long)(q0 + row) * Hq * D + c);
        r.dd[it] = *reinterpret_cast<const shortx8*>(
            dout + q_base + (long)(q0 + row) * Hq * D + c);
      } else {
#pragma unroll
        for (int m = 0; m < 8; ++m) { r.qq[it][m] = 0; r.dd[it][m] = 0; }
      }
    }
    if (tid < TQ) {
      r.stat = (q0 + tid < Tq) ? lse_h[q0 + tid] * 1.44269504f : 0.f;
    } else if (DO_DK && tid < 2 * TQ) {
      r.stat = (q0 + tid - TQ < Tq) ? delta_h[q0 + tid - TQ] : 0.f;
    }
  }

  static __device__ void stage_write(const Stage& r, char* smem_raw) {
    unsigned short* base = reinterpret_cast<unsigned short*>(smem_raw);
    float* stats = reinterpret_cast<float*>(smem_raw + O_STATS);
    const int tid = threadIdx.x;
#pragma unroll
    for (int it = 0; it < NCH; ++it) {
      const int i = tid + it * NTV;
      if (NCHS % NTV != 0 && i >= NCHS) continue;
      const int row = (i * 8) / D, c = (i * 8) % D;
      const int sub = row / QBLK, rin = row % QBLK;
      *reinterpret_cast<shortx8*>(
          &base[O_QROW + sub * (QBLK * DP) + rin * DP + c]) = r.qq[it];
      const int pan = sub * PANEL + (c >> 4) * (QBLK * 16) + rin * 16
          + (c & 15);
      if constexpr (DO_DK) {
        *reinterpret_cast<shortx8*>(
            &base[O_DOROW + sub * (QBLK * DP) + rin * DP + c]) = r.dd[it];
        *reinterpret_cast<shortx8*>(&base[O_QT + pan]) = r.qq[it];
      }
      if constexpr (DO_DV) {
        *reinterpret_cast<shortx8*>(&base[O_DOT + pan]) = r.dd[it];
      }
    }
    if (tid < TQ) stats[tid] = r.stat;
    else if (DO_DK && tid < 2 * TQ) stats[tid] = r.stat;
  }

  static __device__ void run(const unsigned short* q, const unsigned short* k,
                             const unsigned short* v, long vp,
                             const unsigned short* dout,
                             const float* lse, const float* delta,
                             unsigned short* dk, unsigned short* dv,
                             int B, int Tq, int Tkv, int q_off, int Hq,
                             int Hkv, float scale, long dvp, long dvc,
                             char* smem_raw) {
    unsigned short* lds = reinterpret_cast<unsigned short*>(smem_raw);
    float* lse_s = reinterpret_cast<float*>(smem_raw + O_STATS);
    float* delta_s = lse_s + TQ;
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int hi = lane >> 5, ln31 = lane & 31;
    const int hkv = blockIdx.y, b = blockIdx.z;
    const int rep = Hq / Hkv;
    const int kvblk0 = blockIdx.x * WG_KV;  // heavy-first is natural here
    const int wv_kv0 = kvblk0 + wid * QBLK;
    const int kvg = wv_kv0 + ln31;          // this lane's kv row
    const float scale2k = scale * 1.44269504f;

    const long kv_base = (((long)b * Tkv) * Hkv + hkv) * D;
    const long v_base = ((long)b * Tkv) * vp + (long)hkv * D;

    if constexpr (D != DS) {
#pragma unroll
      for (int bufi = 0; bufi < (DBUF ? 2 : 1); ++bufi) {
        unsigned short* lz = reinterpret_cast<unsigned short*>(
            smem_raw + bufi * BUF_BYTES);
#pragma unroll
        for (int sub = 0; sub < NSUB; ++sub)
          for (int i = threadIdx.x; i < (DS / 16 - D / 16) * QBLK * 16;
               i += NTV) {
            const int off = sub * PANEL + (D / 16) * QBLK * 16 + i;
            if constexpr (DO_DK) lz[O_QT + off] = 0;
            if constexpr (DO_DV) lz[O_DOT + off] = 0;
          }
      }
    }

    bf16x8 kfrag[NDSTEP], vfrag[DO_DK ? NDSTEP : 1];
    {
      const unsigned short* kr = k + kv_base + (long)kvg * Hkv * D;
      const unsigned short* vr = v + v_base + (long)kvg * vp;
#pragma unroll
      for (int s = 0; s < NDSTEP; ++s) {
        if (kvg < Tkv) {
          kfrag[s] = *reinterpret_cast<const bf16x8*>(kr + hi * 8 + 16 * s);
          if constexpr (DO_DK)
            vfrag[s] = *reinterpret_cast<const bf16x8*>(vr + hi * 8 + 16 * s);
        } else {
#pragma unroll
          for (int m = 0; m < 8; ++m) {
            kfrag[s][m] = (__bf16)0.f;
            if constexpr (DO_DK) vfrag[s][m] = (__bf16)0.f;
          }
        }
      }
    }

    floatx16 acc_dk[DO_DK ? NDBLK : 1], acc_dv[DO_DV ? NDBLK : 1];
#pragma unroll
    for (int dblk = 0; dblk < NDBLK; ++dblk)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        if constexpr (DO_DK) acc_dk[dblk][r] = 0.f;
        if constexpr (DO_DV) acc_dv[dblk][r] = 0.f;
      }

    // flattened (GQA q-head) x (q stage of TQ rows) iteration space with
    // incremental cursors (div/mod-free; see v1)
    const int first_qtile = max(0, kvblk0 - q_off) / TQ;
    const int n_qtiles = (Tq + TQ - 1) / TQ;
    const int tiles_per_head = n_qtiles - first_qtile;
    const int n_iter = rep * tiles_per_head;
    const int q0_first = first_qtile * TQ;

    // Cursors are two small ints (q0, GQA head index); the pointer bases
    // are recomputed at each stage_load — a handful of VALU once per
    // stage instead of 6 live registers per cursor (this kernel sits at
    // the 256-register line).
    struct Cursor {
      int q0, hq;
    };
    auto cursor_init = [&](Cursor& c) {
      c.q0 = q0_first;
      c.hq = hkv * rep;
    };
    auto cursor_next = [&](Cursor& c) {
      c.q0 += TQ;
      if (c.q0 >= n_qtiles * TQ) {
        c.q0 = q0_first;
        c.hq += 1;
      }
    };
    auto do_stage_load = [&](Stage& st_, const Cursor& c) {
      const long qb = (((long)b * Tq) * Hq + c.hq) * D;
      const float* lh = lse + (((long)b * Hq) + c.hq) * Tq;
      const float* dh = delta + (((long)b * Hq) + c.hq) * Tq;
      stage_load(st_, q, dout, qb, lh, dh, c.q0, Tq, Hq);
    };

    Stage st;
    Cursor cc, cp;
    cursor_init(cc);
    cursor_init(cp);
    do_stage_load(st, cc);
    stage_write(st, smem_raw);
    __syncthreads();
    if (n_iter > 1) {
      cursor_next(cp);
      do_stage_load(st, cp);
    }

    const int lam = ln31 & 15;
    const unsigned tr_lane = (unsigned)(ln31 >> 4) * (QBLK * 16 * 2)
        + (unsigned)hi * 256 + ((lam >> 2) * 16 + 4 * (lam & 3)) * 2;
    const unsigned qt_trb0 = lds_addr(lds + O_QT) + tr_lane;
    const unsigned dot_trb0 = lds_addr(lds + O_DOT) + tr_lane;


    for (int t = 0; t < n_iter; ++t) {
      const int q0t = cc.q0;
      cursor_next(cc);
      const unsigned boff = DBUF ? (unsigned)(t & 1) * BUF_BYTES : 0u;
      unsigned short* const ldsb =
          reinterpret_cast<unsigned short*>(smem_raw + boff);
      const float* const lse_b =
          reinterpret_cast<const float*>(smem_raw + boff + O_STATS);
      const float* const delta_b = lse_b + TQ;
      const unsigned qt_trb = qt_trb0 + boff;
      const unsigned dot_trb = dot_trb0 + boff;
      if constexpr (DBUF) {
        // single-barrier T14: stage t+1 into the other buffer and issue
        // t+2's loads BEFORE this tile's compute
        if (t + 1 < n_iter) {
          stage_write(st, smem_raw + (unsigned)(~t & 1) * BUF_BYTES);
          if (t + 2 < n_iter) {
            cursor_next(cp);
            do_stage_load(st, cp);
          }
        }
      }
      // a stage strictly below this wave's kv rows is fully masked for it
      const bool stage_active = (q0t + TQ - 1 + q_off >= wv_kv0);

      if (stage_active) attnc::static_for<NSUB>([&](auto sub_) {
        constexpr int sub = decltype(sub_)::value;
        const int q0 = q0t + sub * QBLK;
        if (q0 + QBLK - 1 + q_off < wv_kv0) return;  // sub fully masked
        const float* lse_sub = lse_b + sub * 32;
        const float* delta_sub = delta_b + sub * 32;
        // S[q][kv], dP[q][kv] (col = kv = ln31, rows q = crow)
        floatx16 s_acc, dp_acc;
#pragma unroll
        for (int r = 0; r < 16; ++r) { s_acc[r] = 0.f; dp_acc[r] = 0.f; }
#pragma unroll
        for (int s = 0; s < NDSTEP; ++s) {
          const int col = hi * 8 + 16 * s;
          bf16x8 qa = *reinterpret_cast<const bf16x8*>(
              &ldsb[O_QROW + sub * (QBLK * DP) + ln31 * DP + col]);
          s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, kfrag[s], s_acc,
                                                          0, 0, 0);
          if constexpr (DO_DK) {
            bf16x8 da = *reinterpret_cast<const bf16x8*>(
                &ldsb[O_DOROW + sub * (QBLK * DP) + ln31 * DP + col]);
            dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, vfrag[s],
                                                             dp_acc, 0, 0, 0);
          }
        }

        // softmax back into the accumulator vectors (register-frugal:
        // s_acc becomes P; dp_acc becomes dS' when dK is computed)
        if (q0 + q_off >= wv_kv0 + QBLK && q0 + QBLK <= Tq
            && wv_kv0 + QBLK <= Tkv) {
          // interior: every q row of this sub covers every kv row here
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const float lq = lse_sub[crow(r, hi)];
            const float pv = __builtin_amdgcn_exp2f(
                fmaf(s_acc[r], scale2k, -lq));
            s_acc[r] = pv;
            if constexpr (DO_DK) {
              const float dl = delta_sub[crow(r, hi)];
              dp_acc[r] = scale * pv * (dp_acc[r] - dl);
            }
          }
        } else {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int qg = q0 + crow(r, hi);  // local
            const bool ok = (qg < Tq) && (qg + q_off >= kvg) && (kvg < Tkv);
            const float lq = lse_sub[crow(r, hi)];
            const float pv = ok
                ? __builtin_amdgcn_exp2f(fmaf(s_acc[r], scale2k, -lq)) : 0.f;
            s_acc[r] = pv;
            if constexpr (DO_DK) {
              const float dl = delta_sub[crow(r, hi)];
              dp_acc[r] = scale * pv * (dp_acc[r] - dl);
            }
          }
        }
        bf16x8 pfrag[2], dsfrag[2];
        if constexpr (DO_DV) attnc::c16_to_frags(s_acc, pfrag);
        if constexpr (DO_DK) attnc::c16_to_frags(dp_acc, dsfrag);

        // dV += P^T dO ; dK += dS'^T Q  (B-operands via tr reads; base
        // VGPR + offset immediates, sub block folded into the immediate)
        attnc::static_for<NDBLK>([&](auto dblk_) {
          constexpr int dblk = decltype(dblk_)::value;
          auto step = [&](auto s_, uintx2* rd, uintx2* rq) {
            constexpr int s = decltype(s_)::value;
            constexpr unsigned a = sub * (PANEL * 2)
                + dblk * 2 * (QBLK * 16) * 2 + s * 512;
            if constexpr (DO_DV) {
              rd[0] = attnc::tr_read_b64_off<a>(dot_trb);
              rd[1] = attnc::tr_read_b64_off<a + 128>(dot_trb);
            }
            if constexpr (DO_DK) {
              rq[0] = attnc::tr_read_b64_off<a>(qt_trb);
              rq[1] = attnc::tr_read_b64_off<a + 128>(qt_trb);
            }
          };
          auto consume = [&](int s, uintx2* rd, uintx2* rq) {
            if constexpr (DO_DV) {
              unsigned wd[4] = {rd[0][0], rd[0][1], rd[1][0], rd[1][1]};
              bf16x8 dob = *reinterpret_cast<bf16x8*>(wd);
              acc_dv[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                  pfrag[s], dob, acc_dv[dblk], 0, 0, 0);
            }
            if constexpr (DO_DK) {
              unsigned wq[4] = {rq[0][0], rq[0][1], rq[1][0], rq[1][1]};
              bf16x8 qb = *reinterpret_cast<bf16x8*>(wq);
              acc_dk[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                  dsfrag[s], qb, acc_dk[dblk], 0, 0, 0);
            }
          };
          if constexpr (TRHALF) {
            // one k-step per drain (half the in-flight read registers)
            uintx2 rd[2], rq[2];
            attnc::static_for<2>([&](auto s_) {
              step(s_, rd, rq);
              asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
              __builtin_amdgcn_sched_barrier(0);
              consume(decltype(s_)::value, rd, rq);
            });
          } else {
            uintx2 rd[4], rq[4];
            attnc::static_for<2>([&](auto s_) {
              constexpr int s = decltype(s_)::value;
              step(s_, rd + 2 * s, rq + 2 * s);
            });
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_sched_barrier(0);
#pragma unroll
            for (int s = 0; s < 2; ++s) consume(s, rd + 2 * s, rq + 2 * s);
          }
        });
      });  // sub / stage_active

      __syncthreads();
      if constexpr (!DBUF) {
        if (t + 1 < n_iter) {
          stage_write(st, smem_raw);
          if (t + 2 < n_iter) {
            cursor_next(cp);
            do_stage_load(st, cp);
          }
          __syncthreads();
        }
      }
    }

    // stores: col = ln31 = d within block, rows crow = local kv row
    const bool all_in = (kvblk0 + WG_KV <= Tkv);
#pragma unroll
    for (int dblk = 0; dblk < NDBLK; ++dblk) {
      const int d = dblk * 32 + ln31;
      if (D != DS && d >= D) continue;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kvrow = wv_kv0 + crow(r, hi);
        if (all_in || kvrow < Tkv) {
          if constexpr (DO_DK) {
            const long off = kv_base + (long)kvrow * Hkv * D + d;
            union { __bf16 h; unsigned short u; } ck;
            ck.h = (__bf16)acc_dk[dblk][r];
            dk[off] = ck.u;
          }
          if constexpr (DO_DV) {
            const long dvoff = ((long)b * Tkv + kvrow) * dvp + dvc
                + (long)hkv * D + d;
            union { __bf16 h; unsigned short u; } cv;
            cv.h = (__bf16)acc_dv[dblk][r];
            dv[dvoff] = cv.u;
          }
        }
      }
    }
  }
};

// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(NT, 2) void fwd2_kernel(
    const unsigned short* q, const unsigned short* k, const unsigned short* v,
    long vp, unsigned short* o, float* lse, int B, int Tq, int Tkv, int q_off,
    int Hq, int Hkv, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  FwdV2<D>::run(q, k, v, vp, o, lse, B, Tq, Tkv, q_off, Hq, Hkv, scale,
                smem_raw);
}

template <int D>
__global__ __launch_bounds__(NT, 2) void dq2_kernel(
    const unsigned short* q, const unsigned short* k, const unsigned short* v,
    long vp, const unsigned short* dout, const float* lse, const float* delta,
    unsigned short* dq, int B, int Tq, int Tkv, int q_off, int Hq, int Hkv,
    float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  DQV2<D>::run(q, k, v, vp, dout, lse, delta, dq, B, Tq, Tkv, q_off, Hq, Hkv,
               scale, smem_raw);
}

template <int D, int MODE>
__global__ __launch_bounds__((DKDVV2<D, MODE>::NTV),
                             ((MODE == 1 && D == 80) ? 3 : 2))
void dkdv2_kernel(
    const unsigned short* q, const unsigned short* k, const unsigned short* v,
    long vp, const unsigned short* dout, const float* lse, const float* delta,
    unsigned short* dk, unsigned short* dv, int B, int Tq, int Tkv, int q_off,
    int Hq, int Hkv, float scale, long dvp, long dvc) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  DKDVV2<D, MODE>::run(q, k, v, vp, dout, lse, delta, dk, dv, B, Tq, Tkv,
                       q_off, Hq, Hkv, scale, dvp, dvc, smem_raw);
}

// delta preprocess: delta[b,h,t] = rowsum(dO * O) (local copy — device
// code does not link across TUs without rdc)
__global__ void delta2_kernel(const unsigned short* __restrict__ dout,
                              const unsigned short* __restrict__ o,
                              float* __restrict__ delta,
                              long total_rows, int T, int H, int D) {
  const long row = (blockIdx.x * (long)blockDim.x + threadIdx.x) / WAVE_SIZE;
  const int lane = threadIdx.x & 63;
  if (row >= total_rows) return;  // row = ((b*T)+t)*H + h
  const long base = row * D;
  float acc = 0.f;
  for (int d = lane * 2; d < D; d += WAVE_SIZE * 2) {
    unsigned int du = *reinterpret_cast<const unsigned int*>(dout + base + d);
    unsigned int ou = *reinterpret_cast<const unsigned int*>(o + base + d);
    acc += bf16_to_f32((unsigned short)(du & 0xffff)) *
           bf16_to_f32((unsigned short)(ou & 0xffff));
    acc += bf16_to_f32((unsigned short)(du >> 16)) *
           bf16_to_f32((unsigned short)(ou >> 16));
  }
  acc = wave_reduce_sum(acc);
  if (lane == 0) {
    const int h = (int)(row % H);
    const long bt = row / H;
    const int t = (int)(bt % T);
    const long b = bt / T;
    delta[((b * H) + h) * T + t] = acc;
  }
}

}  // namespace attn2

// ---------------------------------------------------------------------------
std::vector<torch::Tensor> attn_fwd2(torch::Tensor q, torch::Tensor k,
                                     torch::Tensor v, bool causal,
                                     long q_offset) {
  TORCH_CHECK(causal, "attn_fwd2: only causal attention is implemented");
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16 && q.dim() == 4
              && q.is_contiguous(), "q must be contiguous bf16 [B,T,Hq,D]");
  const int B = q.size(0), Tq = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Tkv = k.size(1), Hkv = k.size(2);
  // V may be a row-strided view into the joint QKV activation (the fused
  // path reads it in place): strides must be (Tkv*vp, vp, D, 1).
  TORCH_CHECK(k.is_contiguous());
  TORCH_CHECK(v.stride(3) == 1 && v.stride(2) == D
              && (B == 1 || v.stride(0) == (long)Tkv * v.stride(1)),
              "v must be contiguous or a [B,T,Hkv,D] view with row stride");
  // (B == 1: the batch-stride term is never used, so a dim-1 slice of a
  // longer KV-cache buffer is read in place — zero-copy decode)
  const long vp = v.stride(1);
  TORCH_CHECK(Hq % Hkv == 0, "GQA requires Hq % Hkv == 0");
  TORCH_CHECK(D == 64 || D == 80 || D == 128,
              "attn_fwd2: head_dim must be 64, 80 or 128");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, Hq, Tq}, q.options().dtype(torch::kFloat32));
  const float scale = 1.0f / sqrtf((float)D);
  const int n_qblk = (Tq + attn2::QBLK * attn2::NW - 1) / (attn2::QBLK * attn2::NW);
  dim3 grid(n_qblk, Hq, B);
  auto stream = at::cuda::getCurrentHIPStream();
  auto launch = [&](auto kfn, size_t smem) {
    hipLaunchKernelGGL(kfn, grid, dim3(attn2::NT), smem, stream,
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(), vp,
                       (unsigned short*)o.data_ptr(), lse.data_ptr<float>(),
                       B, Tq, Tkv, (int)q_offset, Hq, Hkv, scale);
  };
  if (D == 128) launch(attn2::fwd2_kernel<128>, attn2::FwdV2<128>::SMEM_BYTES);
  else if (D == 80) launch(attn2::fwd2_kernel<80>, attn2::FwdV2<80>::SMEM_BYTES);
  else launch(attn2::fwd2_kernel<64>, attn2::FwdV2<64>::SMEM_BYTES);
  HIP_CHECK_KERNEL();
  return {o, lse};
}

static std::vector<torch::Tensor> attn_bwd2_impl(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor o, torch::Tensor lse, bool causal, long q_offset,
    torch::Tensor dv, long dvp, long dvc) {
  TORCH_CHECK(causal, "attn_bwd2: only causal attention is implemented");
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16 && q.is_contiguous());
  const int B = q.size(0), T = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Tkv = k.size(1), Hkv = k.size(2);
  TORCH_CHECK(v.stride(3) == 1 && v.stride(2) == D
              && (B == 1 || v.stride(0) == (long)Tkv * v.stride(1)),
              "v must be contiguous or a [B,T,Hkv,D] view with row stride");
  // (B == 1: the batch-stride term is never used, so a dim-1 slice of a
  // longer KV-cache buffer is read in place — zero-copy decode)
  const long vstride = v.stride(1);
  const int q_off = (int)q_offset;
  const float scale = 1.0f / sqrtf((float)D);
  TORCH_CHECK(D == 64 || D == 80 || D == 128,
              "attn_bwd2: head_dim must be 64, 80 or 128");
  auto stream = at::cuda::getCurrentHIPStream();

  auto delta = torch::empty({B, Hq, T}, q.options().dtype(torch::kFloat32));
  {
    long total_rows = (long)B * T * Hq;
    long threads = total_rows * WAVE_SIZE;
    int block = 256;
    long grid = (threads + block - 1) / block;
    hipLaunchKernelGGL(attn2::delta2_kernel, dim3((unsigned)grid), dim3(block),
                       0, stream, (const unsigned short*)dout.data_ptr(),
                       (const unsigned short*)o.data_ptr(),
                       delta.data_ptr<float>(), total_rows, T, Hq, D);
    HIP_CHECK_KERNEL();
  }

  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);

  // dq, dv and dk are independent given delta: run on separate streams so
  // their tail waves overlap (raw fork/join events keep this
  // hipGraph-capturable)
  static hipStream_t side = nullptr, side2 = nullptr;
  static hipEvent_t ev_fork = nullptr, ev_join = nullptr, ev_join2 = nullptr;
  if (side == nullptr) {
    hipStreamCreateWithFlags(&side, hipStreamNonBlocking);
    hipStreamCreateWithFlags(&side2, hipStreamNonBlocking);
    hipEventCreateWithFlags(&ev_fork, hipEventDisableTiming);
    hipEventCreateWithFlags(&ev_join, hipEventDisableTiming);
    hipEventCreateWithFlags(&ev_join2, hipEventDisableTiming);
  }
  hipEventRecord(ev_fork, stream);
  hipStreamWaitEvent(side, ev_fork, 0);

  const int WG = attn2::QBLK * attn2::NW;  // 256 q rows per dq workgroup
  const int n_qblk = (T + WG - 1) / WG;
  auto launch_dq = [&](auto dqk, size_t smem_dq) {
    hipLaunchKernelGGL(dqk, dim3(n_qblk, Hq, B), dim3(attn2::NT), smem_dq,
                       stream, (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(), vstride,
                       (const unsigned short*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (unsigned short*)dq.data_ptr(), B, T, Tkv, q_off, Hq,
                       Hkv, scale);
    HIP_CHECK_KERNEL();
  };
  auto launch_kv = [&](auto kfn, int wg_kv, int nthreads, size_t smem,
                       hipStream_t st) {
    const int n_kvblk = (Tkv + wg_kv - 1) / wg_kv;
    hipLaunchKernelGGL(kfn, dim3(n_kvblk, Hkv, B), dim3(nthreads), smem,
                       st, (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(), vstride,
                       (const unsigned short*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (unsigned short*)dk.data_ptr(),
                       (unsigned short*)dv.data_ptr(), B, T, Tkv, q_off, Hq,
                       Hkv, scale, dvp, dvc);
    HIP_CHECK_KERNEL();
  };
  bool used_side2 = false;
  if (D == 128) {
    // fused dkdv needs ~290 registers at D=128: run the register-clean
    // dV-only / dK-only kernels concurrently instead
    using DV = attn2::DKDVV2<128, 1>;
    using DK = attn2::DKDVV2<128, 2>;
    launch_dq(attn2::dq2_kernel<128>, attn2::DQV2<128>::SMEM_BYTES);
    hipStreamWaitEvent(side2, ev_fork, 0);
    used_side2 = true;
    launch_kv(attn2::dkdv2_kernel<128, 1>, DV::WG_KV, DV::NTV, DV::SMEM_BYTES,
              side);
    launch_kv(attn2::dkdv2_kernel<128, 2>, DK::WG_KV, DK::NTV, DK::SMEM_BYTES,
              side2);
  } else if (D == 80) {
    // A/B: fused dkdv (1 kernel, 8 waves, 254 regs) vs split dv/dk
    // (register-light 4-wave kernels, +25% MFMA) — MA_DKDV80_SPLIT=0/1
    static const bool split80 = [] {
      const char* e = getenv("MA_DKDV80_SPLIT");
      return e ? atoi(e) != 0 : false;
    }();
    launch_dq(attn2::dq2_kernel<80>, attn2::DQV2<80>::SMEM_BYTES);
    if (split80) {
      using DV = attn2::DKDVV2<80, 1>;
      using DK = attn2::DKDVV2<80, 2>;
      hipStreamWaitEvent(side2, ev_fork, 0);
      used_side2 = true;
      launch_kv(attn2::dkdv2_kernel<80, 1>, DV::WG_KV, DV::NTV,
                DV::SMEM_BYTES, side);
      launch_kv(attn2::dkdv2_kernel<80, 2>, DK::WG_KV, DK::NTV,
                DK::SMEM_BYTES, side2);
    } else {
      using F = attn2::DKDVV2<80, 0>;
      launch_kv(attn2::dkdv2_kernel<80, 0>, F::WG_KV, F::NTV, F::SMEM_BYTES,
                side);
    }
  } else {
    using F = attn2::DKDVV2<64, 0>;
    launch_dq(attn2::dq2_kernel<64>, attn2::DQV2<64>::SMEM_BYTES);
    launch_kv(attn2::dkdv2_kernel<64, 0>, F::WG_KV, F::NTV, F::SMEM_BYTES,
              side);
  }
  hipEventRecord(ev_join, side);
  hipStreamWaitEvent(stream, ev_join, 0);
  if (used_side2) {
    hipEventRecord(ev_join2, side2);
    hipStreamWaitEvent(stream, ev_join2, 0);
  }
  // dk/dv (and the delta reads) are touched by kernels on the side
  // streams but were allocated on the caller's stream: record the side
  // streams so the caching allocator cannot hand their pages to another
  // stream before the side work completes (join orders only `stream`).
  auto rec = [&](const torch::Tensor& t, hipStream_t st) {
    c10::hip::HIPCachingAllocator::recordStream(
        t.storage().data_ptr(),
        c10::hip::getStreamFromExternal(st, q.get_device()));
  };
  rec(dk, side);
  rec(dv, side);
  rec(delta, side);
  if (used_side2) {
    rec(dk, side2);
    rec(dv, side2);
    rec(delta, side2);
  }
  return {dq, dk, dv};
}

std::vector<torch::Tensor> attn_bwd2(torch::Tensor dout, torch::Tensor q,
                                     torch::Tensor k, torch::Tensor v,
                                     torch::Tensor o, torch::Tensor lse,
                                     bool causal, long q_offset) {
  // dv is always a fresh contiguous tensor (v itself may be a strided view)
  auto dv = torch::empty(v.sizes(), q.options());
  const long Hkv = v.size(2), D = v.size(3);
  auto r = attn_bwd2_impl(dout, q, k, v, o, lse, causal, q_offset, dv,
                          Hkv * D, 0);
  return {r[0], r[1], dv};
}

std::vector<torch::Tensor> attn_bwd2_qkvjoint(torch::Tensor dout,
                                              torch::Tensor q, torch::Tensor k,
                                              torch::Tensor v, torch::Tensor o,
                                              torch::Tensor lse, bool causal,
                                              long q_offset, torch::Tensor dqkv,
                                              long dv_col_off) {
  TORCH_CHECK(dqkv.is_cuda() && dqkv.dtype() == torch::kBFloat16
              && dqkv.dim() == 3 && dqkv.is_contiguous());
  const long Ctot = dqkv.size(2), Hkv = v.size(2), D = v.size(3);
  TORCH_CHECK(dqkv.size(0) == v.size(0) && dqkv.size(1) == v.size(1)
              && dv_col_off + Hkv * D <= Ctot);
  auto r = attn_bwd2_impl(dout, q, k, v, o, lse, causal, q_offset, dqkv,
                          Ctot, dv_col_off);
  return {r[0], r[1]};
}
