// K1 (backward): flash attention backward for CDNA4/gfx950, GQA-aware.
// FA2-style split (no atomics):
//   1. delta[b,h,q]  = rowsum(dO * O)                       (preprocess)
//   2. dQ kernel  : per q-block, iterate kv tiles            (recomputes S)
//   3. dKdV kernel: per kv-block, iterate q heads x q tiles  (recomputes S)
// Math (S' = Q K^T, S = scale*S', P = exp(S - lse)):
//   dV = P^T dO ; dP = dO V^T ; dS' = scale * P .* (dP - delta)
//   dQ = dS' K  ; dK = dS'^T Q
// All matmuls are 32x32x16 bf16 MFMAs using the same fragment maps as the
// forward (A[i=lane&31][k=hi*8+m], B[k=hi*8+m][j=lane&31], D col=lane&31 /
// row=crow(reg,hi)). In-register C->A/B layout transposes go through
// cvt_pk + permlane32_swap (guide T12); transposed Q/dO/K operands come
// from [panel][row][16] LDS images via ds_read_b64_tr_b16 (guide T10);
// both kernels software-pipeline their tile staging (dkdv: double-buffered
// two-deep unroll; dq: T14 issue-early/write-late).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(2))) unsigned int uintx2;

namespace attnbwd {

// ds_read_b64_tr_b16 (semantics verified by tr16_probe on gfx950): the 16
// lanes of each contiguous lane group supply one 8-byte (4x bf16) chunk
// each; with the [panel][row][16-col] image and per-lane address
//   base + (panel*512 + (k0 + (lambda>>2))*16 + 4*(lambda&3)) * 2B,
// lambda = lane&15, the read returns to lane l element j = X[k0+j][d0+(l&15)]
// -- exactly the 32x32x16 MFMA A/B fragment k-run (4 of the 8 elements;
// two reads at k0 and k0+4 complete it). Conflict-free: the 16 lanes hit 16
// distinct even dwords of a 256-dword-aligned panel.
__device__ __forceinline__ uintx2 tr_read_b64(unsigned addr_bytes) {
  uintx2 r;
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(r) : "v"(addr_bytes));
  return r;
}

__device__ __forceinline__ unsigned lds_addr(const void* p) {
  return (unsigned)(unsigned long long)p;  // LDS aperture: low 32 = offset
}

constexpr int KVBLK = 64;  // dQ kernel kv tile

// Swizzles (see attention_fwd.hip for the derivation):
__device__ __forceinline__ int swz(int row, int col) {   // row-major tiles
  return col ^ (((row & 3) | (((row >> 4) & 1) << 2)) << 3);
}
__device__ __forceinline__ int crow(int reg, int hi) {
  return (reg & 3) + 8 * (reg >> 2) + 4 * hi;
}

// Re-layout f32 values from the MFMA C layout (col = lane&31 = j fixed,
// rows i = crow(reg,hi)) into bf16 fragments with per-lane element
// k = hi*8+m (+16 per frag). Used as operand A the result represents
// X^T[j=lane&31][k=i]: exactly the transpose-in-place the bwd needs.
// nfrag = source_rows/16 (regs p[8s..8s+7] cover rows 16s..16s+15).
__device__ __forceinline__ void c_layout_to_frags(const float* p, bf16x8* frag,
                                                  int nfrag) {
  for (int s = 0; s < nfrag; ++s) {
    unsigned int w[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      w[i] = ((unsigned int)f32_to_bf16(p[8 * s + 2 * i])) |
             ((unsigned int)f32_to_bf16(p[8 * s + 2 * i + 1]) << 16);
    }
    auto r02 = __builtin_amdgcn_permlane32_swap(w[0], w[2], false, false);
    auto r13 = __builtin_amdgcn_permlane32_swap(w[1], w[3], false, false);
    unsigned int fw[4] = {(unsigned int)r02[0], (unsigned int)r13[0],
                          (unsigned int)r02[1], (unsigned int)r13[1]};
    frag[s] = *reinterpret_cast<bf16x8*>(fw);
  }
}

// ---------------- delta preprocess ----------------------------------------
__global__ void delta_kernel(const unsigned short* __restrict__ dout,
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

// ---------------- dQ kernel -----------------------------------------------
// 4 waves x 32 q rows = 128 q rows per WG; kv tiles of 64 in LDS (K row +
// V row as S^T/dP^T A-operands; Kt as the dQ^T A-operand). S^T/P^T/dS'^T
// all live in the col = q = lane&31 layout (like the forward), so lse and
// delta are lane-local.
template <int D>
struct DQKernel {
  static constexpr int NDSTEP = D / 16;
  static constexpr int NDBLK = D / 32;
  struct Smem {
    unsigned short k[KVBLK * D];
    unsigned short v[KVBLK * D];
    unsigned short kt[D * KVBLK];  // panel image [D/16][KVBLK][16] for tr reads
  };

  static __device__ void run(const unsigned short* q, const unsigned short* k,
                             const unsigned short* v, const unsigned short* dout,
                             const float* lse, const float* delta,
                             unsigned short* dq, int B, int Tq, int Tkv,
                             int q_off, int Hq, int Hkv, float scale,
                             char* smem_raw) {
    Smem* sm = reinterpret_cast<Smem*>(smem_raw);
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int hi = lane >> 5, ln31 = lane & 31;
    const int h = blockIdx.y, b = blockIdx.z;
    const int hkv = h / (Hq / Hkv);
    const int qblk0 = blockIdx.x * 128;
    const int qg = qblk0 + wid * 32 + ln31;   // LOCAL q row
    const int qgl = qg + q_off;               // global position vs keys

    const long q_base = (((long)b * Tq) * Hq + h) * D;
    const long kv_base = (((long)b * Tkv) * Hkv + hkv) * D;

    bf16x8 qfrag[NDSTEP], dofrag[NDSTEP];
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
    // log2-domain lse (exp2 fold; see attention_fwd.hip)
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

    const int q_hi_wg = min(qblk0 + 127, Tq - 1) + q_off;  // global kv bound
    const int n_tiles = min(q_hi_wg, Tkv - 1) / KVBLK + 1;

    // T14 issue-early/write-late with ONE register set and ONE LDS buffer:
    // the next tile's global loads are issued before this tile's compute
    // (HBM latency hides under the MFMAs); the LDS write happens after the
    // barrier that ends this tile's reads. Keeps the kernel at 2 waves/SIMD
    // (a second buffer or register set would cross the 256-VGPR cliff).
    constexpr int NCH = KVBLK * D / 8 / 256;  // chunks per thread
    shortx8 kreg[NCH], vreg[NCH];
    const int tid = threadIdx.x;

    auto stage_load = [&](int kv0) {
#pragma unroll
      for (int it = 0; it < NCH; ++it) {
        const int i = tid + it * 256;
        const int r = (i * 8) / D, c = (i * 8) % D;
        if (kv0 + r < Tkv) {
          kreg[it] = *reinterpret_cast<const shortx8*>(
              k + kv_base + (long)(kv0 + r) * Hkv * D + c);
          vreg[it] = *reinterpret_cast<const shortx8*>(
              v + kv_base + (long)(kv0 + r) * Hkv * D + c);
        } else {
#pragma unroll
          for (int m = 0; m < 8; ++m) { kreg[it][m] = 0; vreg[it][m] = 0; }
        }
      }
    };
    auto stage_write = [&]() {
#pragma unroll
      for (int it = 0; it < NCH; ++it) {
        const int i = tid + it * 256;
        const int r = (i * 8) / D, c = (i * 8) % D;
        *reinterpret_cast<shortx8*>(&sm->k[r * D + swz(r, c)]) = kreg[it];
        *reinterpret_cast<shortx8*>(&sm->v[r * D + swz(r, c)]) = vreg[it];
        *reinterpret_cast<shortx8*>(
            &sm->kt[(c >> 4) * (KVBLK * 16) + r * 16 + (c & 15)]) = kreg[it];
      }
    };

    stage_load(0);
    stage_write();
    if (n_tiles > 1) stage_load(KVBLK);
    __syncthreads();

    for (int tile = 0; tile < n_tiles; ++tile) {
      const int kv0 = tile * KVBLK;

      // S^T = K Q^T ; dP^T = V dO^T   (both: col = q = ln31, rows = kv)
      floatx16 s0, s1, dp0, dp1;
#pragma unroll
      for (int r = 0; r < 16; ++r) { s0[r] = s1[r] = dp0[r] = dp1[r] = 0.f; }
#pragma unroll
      for (int s = 0; s < NDSTEP; ++s) {
        const int col = hi * 8 + 16 * s;
        const int r0 = ln31, r1 = ln31 + 32;
        bf16x8 ka = *reinterpret_cast<const bf16x8*>(&sm->k[r0 * D + swz(r0, col)]);
        bf16x8 kb = *reinterpret_cast<const bf16x8*>(&sm->k[r1 * D + swz(r1, col)]);
        bf16x8 va = *reinterpret_cast<const bf16x8*>(&sm->v[r0 * D + swz(r0, col)]);
        bf16x8 vb = *reinterpret_cast<const bf16x8*>(&sm->v[r1 * D + swz(r1, col)]);
        s0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[s], s0, 0, 0, 0);
        s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kb, qfrag[s], s1, 0, 0, 0);
        dp0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, dofrag[s], dp0, 0, 0, 0);
        dp1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vb, dofrag[s], dp1, 0, 0, 0);
      }

      // dS'^T = scale * P .* (dP - delta), P = exp(scale*S' - lse)
      float ds[32];
      // interior fast path: the whole kv tile is strictly below this
      // wave's lowest (global) q row -> skip the causal/range compares
      // (VALU-issue-bound, like the forward)
      const int q_lo_wave = qblk0 + wid * 32 + q_off;
      if (kv0 + KVBLK - 1 <= q_lo_wave && kv0 + KVBLK <= Tkv && qg < Tq) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float p0 = __builtin_amdgcn_exp2f(s0[r] * scale2 - my_lse2);
          const float p1 = __builtin_amdgcn_exp2f(s1[r] * scale2 - my_lse2);
          ds[r] = scale * p0 * (dp0[r] - my_delta);
          ds[16 + r] = scale * p1 * (dp1[r] - my_delta);
        }
      } else {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kg0 = kv0 + crow(r, hi);
        const int kg1 = kv0 + 32 + crow(r, hi);
        const float p0 = (kg0 <= qgl && kg0 < Tkv)
                             ? __builtin_amdgcn_exp2f(s0[r] * scale2 - my_lse2) : 0.f;
        const float p1 = (kg1 <= qgl && kg1 < Tkv)
                             ? __builtin_amdgcn_exp2f(s1[r] * scale2 - my_lse2) : 0.f;
        ds[r] = scale * p0 * (dp0[r] - my_delta);
        ds[16 + r] = scale * p1 * (dp1[r] - my_delta);
      }
      }
      bf16x8 dsfrag[KVBLK / 16];
      c_layout_to_frags(ds, dsfrag, KVBLK / 16);

      // dQ^T += Kt dS'^T-frags : D[m=d][n=q], col = q = ln31. The Kt
      // A-operand (lane holds K[kv-run][own d]) comes from the panel image
      // via hardware transpose reads (see tr_read_b64).
      {
        const unsigned kt_base = lds_addr(sm->kt);
        const int lam = ln31 & 15;
        const unsigned lane_off = ((lam >> 2) * 16 + 4 * (lam & 3)) * 2;
#pragma unroll
        for (int dblk = 0; dblk < NDBLK; ++dblk) {
          const unsigned pan = (unsigned)((dblk * 2 + (ln31 >> 4))
                                          * (KVBLK * 16)) * 2;
          uintx2 rk[2 * (KVBLK / 16)];
#pragma unroll
          for (int s = 0; s < KVBLK / 16; ++s) {
            const unsigned a = pan + (16 * s + 8 * hi) * 32 + lane_off;
            rk[2 * s] = tr_read_b64(kt_base + a);
            rk[2 * s + 1] = tr_read_b64(kt_base + a + 128);
          }
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
#pragma unroll
          for (int s = 0; s < KVBLK / 16; ++s) {
            unsigned w[4] = {rk[2 * s][0], rk[2 * s][1],
                             rk[2 * s + 1][0], rk[2 * s + 1][1]};
            bf16x8 kta = *reinterpret_cast<bf16x8*>(w);
            acc_dq[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                kta, dsfrag[s], acc_dq[dblk], 0, 0, 0);
          }
        }
      }
      __syncthreads();  // all waves finished reading tile `tile`
      if (tile + 1 < n_tiles) {
        stage_write();                                   // tile+1 -> LDS
        if (tile + 2 < n_tiles) stage_load((tile + 2) * KVBLK);
        __syncthreads();                                 // tile+1 ready
      }
    }

    if (qg < Tq) {
      unsigned short* dqr = dq + q_base + (long)qg * Hq * D;
#pragma unroll
      for (int dblk = 0; dblk < NDBLK; ++dblk)
#pragma unroll
        for (int r4 = 0; r4 < 4; ++r4) {  // pack 4 consecutive d per store
          unsigned int w[2];
#pragma unroll
          for (int i = 0; i < 2; ++i) {
            const int r = r4 * 4 + 2 * i;
            w[i] = (unsigned int)f32_to_bf16(acc_dq[dblk][r]) |
                   ((unsigned int)f32_to_bf16(acc_dq[dblk][r + 1]) << 16);
          }
          const int d = dblk * 32 + crow(r4 * 4, hi);
          *reinterpret_cast<uint2*>(dqr + d) = make_uint2(w[0], w[1]);
        }
    }
  }
};

// ---------------- dKdV kernel ---------------------------------------------
// 4 waves x 32 kv rows = 128 kv rows per WG; loop (q head in GQA group) x
// (q tiles of 32). K,V live in REGISTERS (lane owns kv row
// wavebase+ln31); LDS holds the q tile 4 ways: Q row-major, dO row-major
// (A-operands of S/dP), Qt, dOt transposed (B-operands of dK/dV).
//   S [q][kv]  = mfma(A=Q_lds,  B=K_regs)    col = kv = ln31, rows q = crow
//   dP[q][kv]  = mfma(A=dO_lds, B=V_regs)    same layout
//   dS' = scale * P .* (dP - delta[q])       (lse/delta loaded per q row)
//   dV[kv][d] += mfma(A=frags(P),   B=dOt)   col = d = ln31, rows kv = crow
//   dK[kv][d] += mfma(A=frags(dS'), B=Qt)
// ABL (ablation; guide mistake #8): 0 full; 1 skip transposed qt/dot image
// writes; 2 skip ALL LDS staging writes; 3 skip lse/delta scalar loads;
// 4 skip dV/dK MFMAs; 5 skip S/dP MFMAs + softmax (staging only).
// Modes >0 give WRONG results; timing-only.
template <int D, int ABL = 0>
struct DKDVKernel {
  static constexpr int NDSTEP = D / 16;
  static constexpr int NDBLK = D / 32;
  // Double-buffered q-tile: Q/dO row-major swizzled (S/dP A-operands),
  // Q^T/dO^T transposed swizzled (dK/dV B-operands), and the per-row
  // softmax stats staged once per tile (read back as LDS broadcasts —
  // the v1 per-reg scalar GLOBAL loads of lse/delta were 26% of kernel
  // time by ablation).
  struct Tile {
    unsigned short qrow[32 * D];
    unsigned short dorow[32 * D];
    unsigned short qt[D * 32];
    unsigned short dot[D * 32];
    float lse[32];
    float delta[32];
  };
  struct Smem {
    Tile t[2];
  };

  // One thread's staged registers for a tile (NCH chunks x 8 bf16 x {q,do};
  // 32*D/8 chunks over 256 threads: D=128 -> 2/thread, D=64 -> 1).
  static constexpr int NCH = (32 * D / 8 + 255) / 256;
  struct StageRegs {
    shortx8 qq[NCH], dd[NCH];
    float stat;  // thread's lse/delta element (tid < 64)
  };

  static __device__ void stage_load(StageRegs& r, const unsigned short* q,
                                    const unsigned short* dout, long q_base,
                                    const float* lse_h, const float* delta_h,
                                    int q0, int Tq, int Hq) {
    const int tid = threadIdx.x;
#pragma unroll
    for (int it = 0; it < NCH; ++it) {
      const int i = tid + it * 256;
      const int row = (i * 8) / D, c = (i * 8) % D;
      if (row >= 32) continue;
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
    if (tid < 32) {
      // staged in the log2 domain (exp2 fold) - free at staging time
      r.stat = (q0 + tid < Tq) ? lse_h[q0 + tid] * 1.44269504f : 0.f;
    } else if (tid < 64) {
      r.stat = (q0 + tid - 32 < Tq) ? delta_h[q0 + tid - 32] : 0.f;
    }
  }

  static __device__ void stage_write(const StageRegs& r, Tile* tl) {
    const int tid = threadIdx.x;
#pragma unroll
    for (int it = 0; it < NCH; ++it) {
      const int i = tid + it * 256;
      const int row = (i * 8) / D, c = (i * 8) % D;
      if (row >= 32) continue;
      if constexpr (ABL < 2) {
        *reinterpret_cast<shortx8*>(&tl->qrow[row * D + swz(row, c)]) = r.qq[it];
        *reinterpret_cast<shortx8*>(&tl->dorow[row * D + swz(row, c)]) = r.dd[it];
      } else {
        asm volatile("" :: "v"(r.qq[it][0]), "v"(r.dd[it][0]));
      }
      if constexpr (ABL < 1) {
        // panel image [D/16][32 q][16 d]: the thread's 8 consecutive d of
        // one q row land contiguously -> one 16B write per tensor (was 8
        // scalar b16 scatter stores with 8-way write-bank conflicts)
        const int pan = (c >> 4) * 512 + row * 16 + (c & 15);
        *reinterpret_cast<shortx8*>(&tl->qt[pan]) = r.qq[it];
        *reinterpret_cast<shortx8*>(&tl->dot[pan]) = r.dd[it];
      }
    }
    if (tid < 32) tl->lse[tid] = r.stat;
    else if (tid < 64) tl->delta[tid - 32] = r.stat;
  }

  // dvp/dvc: row pitch (elements per (b,t) row) and column offset for the
  // dv store — lets dv land directly in a fused-QKV gradient buffer
  // [B,T,Cq+Ck+Cv] instead of its own tensor (kills the backward cat).
  static __device__ void run(const unsigned short* q, const unsigned short* k,
                             const unsigned short* v, const unsigned short* dout,
                             const float* lse, const float* delta,
                             unsigned short* dk, unsigned short* dv,
                             int B, int Tq, int Tkv, int q_off, int Hq,
                             int Hkv, float scale, long dvp, long dvc,
                             char* smem_raw) {
    Smem* sm = reinterpret_cast<Smem*>(smem_raw);
    const int lane = threadIdx.x & 63;
    const int wid = threadIdx.x >> 6;
    const int hi = lane >> 5, ln31 = lane & 31;
    const int hkv = blockIdx.y, b = blockIdx.z;
    const int rep = Hq / Hkv;
    const int kvblk0 = blockIdx.x * 128;
    const int kvg = kvblk0 + wid * 32 + ln31;  // this lane's kv row
    const float scale2k = scale * 1.44269504f;  // exp2 fold

    const long kv_base = (((long)b * Tkv) * Hkv + hkv) * D;

    bf16x8 kfrag[NDSTEP], vfrag[NDSTEP];
    {
      const unsigned short* kr = k + kv_base + (long)kvg * Hkv * D;
      const unsigned short* vr = v + kv_base + (long)kvg * Hkv * D;
#pragma unroll
      for (int s = 0; s < NDSTEP; ++s) {
        if (kvg < Tkv) {
          kfrag[s] = *reinterpret_cast<const bf16x8*>(kr + hi * 8 + 16 * s);
          vfrag[s] = *reinterpret_cast<const bf16x8*>(vr + hi * 8 + 16 * s);
        } else {
#pragma unroll
          for (int m = 0; m < 8; ++m) { kfrag[s][m] = (__bf16)0.f;
                                        vfrag[s][m] = (__bf16)0.f; }
        }
      }
    }

    floatx16 acc_dk[NDBLK], acc_dv[NDBLK];
#pragma unroll
    for (int dblk = 0; dblk < NDBLK; ++dblk)
#pragma unroll
      for (int r = 0; r < 16; ++r) { acc_dk[dblk][r] = 0.f; acc_dv[dblk][r] = 0.f; }

    // Flattened (hq, qtile) iteration space, software-pipelined two deep:
    // iter t computes tile t from LDS buf[t&1] while tile t+1's staged
    // registers are written to buf[(t+1)&1] and tile t+2's global loads are
    // issued (T14 issue-early/write-late; one barrier per iteration).
    // causal in LOCAL q coords: need q_local >= kv - q_off
    const int first_qtile = max(0, kvblk0 - q_off) / 32;
    const int n_qtiles = (Tq + 31) / 32;
    const int tiles_per_head = n_qtiles - first_qtile;
    const int n_iter = rep * tiles_per_head;
    const int q0_first = first_qtile * 32;

    // Incremental cursors instead of per-iteration div/mod (an integer
    // division expands to ~25 VALU ops; called 2x per tile it was ~6% of
    // the issue budget). A cursor = {q0, q_base, lse/delta base} advanced
    // by one tile per step, wrapping to the next GQA q-head.
    struct Cursor {
      int q0;
      long q_base;
      const float *lse_h, *delta_h;
    };
    auto cursor_init = [&](Cursor& c) {
      const int hq = hkv * rep;
      c.q0 = q0_first;
      c.q_base = (((long)b * Tq) * Hq + hq) * D;
      c.lse_h = lse + (((long)b * Hq) + hq) * Tq;
      c.delta_h = delta + (((long)b * Hq) + hq) * Tq;
    };
    auto cursor_next = [&](Cursor& c) {
      c.q0 += 32;
      if (c.q0 >= n_qtiles * 32) {  // next q-head of the GQA group
        c.q0 = q0_first;
        c.q_base += (long)D;
        c.lse_h += Tq;
        c.delta_h += Tq;
      }
    };

    StageRegs rA, rB;  // ping-pong staging registers (rule #20: static names)
    Cursor cc, cp;     // compute cursor (tile t), prefetch cursor (tile t+2)
    cursor_init(cc);
    cursor_init(cp);
    {
      stage_load(rA, q, dout, cc.q_base, cc.lse_h, cc.delta_h, cc.q0, Tq, Hq);
      stage_write(rA, &sm->t[0]);
      if (n_iter > 1) {
        cursor_next(cp);
        stage_load(rA, q, dout, cp.q_base, cp.lse_h, cp.delta_h, cp.q0, Tq, Hq);
      }
      __syncthreads();
    }

    // Two-deep unrolled pipeline: the (wr, ld) register-set roles alternate
    // between rA and rB per iteration, eliminating the per-tile struct copy
    // (~34 v_movs) a single rotating set would need.
    auto iter_body = [&](int t, StageRegs& wr, StageRegs& ld) {
      Tile* cur = &sm->t[t & 1];
      // write tile t+1 (in wr) into the other buffer; issue tile t+2 loads
      if (t + 1 < n_iter) {
        stage_write(wr, &sm->t[(t + 1) & 1]);
        if (t + 2 < n_iter) {
          cursor_next(cp);
          stage_load(ld, q, dout, cp.q_base, cp.lse_h, cp.delta_h, cp.q0, Tq, Hq);
        }
      }
      const int q0 = cc.q0;
      cursor_next(cc);

      // S[q][kv] and dP[q][kv] (col = kv = ln31, rows q = crow)
      floatx16 s_acc, dp_acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) { s_acc[r] = 0.f; dp_acc[r] = 0.f; }
      if constexpr (ABL != 5) {
#pragma unroll
        for (int s = 0; s < NDSTEP; ++s) {
          const int col = hi * 8 + 16 * s;
          bf16x8 qa = *reinterpret_cast<const bf16x8*>(
              &cur->qrow[ln31 * D + swz(ln31, col)]);
          bf16x8 da = *reinterpret_cast<const bf16x8*>(
              &cur->dorow[ln31 * D + swz(ln31, col)]);
          s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, kfrag[s], s_acc,
                                                          0, 0, 0);
          dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, vfrag[s], dp_acc,
                                                           0, 0, 0);
        }
      }

      float p_c[16], ds_c[16];
      if constexpr (ABL != 5) {
        if (q0 + q_off >= kvblk0 + 128 && q0 + 32 <= Tq) {
          // interior tile: every q row of this tile is > every kv row of
          // the workgroup and in-range -> no mask compares needed
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const float lq = (ABL != 3) ? cur->lse[crow(r, hi)] : 0.f;
            const float dl = (ABL != 3) ? cur->delta[crow(r, hi)] : 0.f;
            const float pv = __builtin_amdgcn_exp2f(s_acc[r] * scale2k - lq);
            p_c[r] = pv;
            ds_c[r] = scale * pv * (dp_acc[r] - dl);
          }
        } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qg = q0 + crow(r, hi);       // local
          const bool ok = (qg < Tq) && (qg + q_off >= kvg) && (kvg < Tkv);
          // LDS broadcast (same address across the 32 lanes of one half)
          const float lq = (ABL != 3) ? cur->lse[crow(r, hi)] : 0.f;
          const float dl = (ABL != 3) ? cur->delta[crow(r, hi)] : 0.f;
          const float pv = ok ? __builtin_amdgcn_exp2f(s_acc[r] * scale2k - lq) : 0.f;
          p_c[r] = pv;
          ds_c[r] = scale * pv * (dp_acc[r] - dl);
        }
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) { p_c[r] = 0.f; ds_c[r] = 0.f;
          asm volatile("" :: "v"(p_c[r]), "v"(ds_c[r])); }
      }
      bf16x8 pfrag[2], dsfrag[2];
      c_layout_to_frags(p_c, pfrag, 2);
      c_layout_to_frags(ds_c, dsfrag, 2);

      // dV += P^T dO ; dK += dS'^T Q  (B-operands via hardware transpose
      // reads from the panel images; see tr_read_b64 note above)
      if constexpr (ABL != 4) {
        const unsigned qt_base = lds_addr(cur->qt);
        const unsigned dot_base = lds_addr(cur->dot);
        const int lam = ln31 & 15;
        const unsigned lane_off = ((lam >> 2) * 16 + 4 * (lam & 3)) * 2;
#pragma unroll
        for (int dblk = 0; dblk < NDBLK; ++dblk) {
          const unsigned pan = (unsigned)((dblk * 2 + (ln31 >> 4)) * 512) * 2;
          uintx2 rd[4], rq[4];
#pragma unroll
          for (int s = 0; s < 2; ++s) {
            const unsigned k0 = (16 * s + 8 * hi) * 32;  // *16 elems *2 B
            const unsigned a = pan + k0 + lane_off;
            rd[2 * s] = tr_read_b64(dot_base + a);
            rd[2 * s + 1] = tr_read_b64(dot_base + a + 128);  // k0+4 rows
            rq[2 * s] = tr_read_b64(qt_base + a);
            rq[2 * s + 1] = tr_read_b64(qt_base + a + 128);
          }
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
#pragma unroll
          for (int s = 0; s < 2; ++s) {
            unsigned wd[4] = {rd[2 * s][0], rd[2 * s][1],
                              rd[2 * s + 1][0], rd[2 * s + 1][1]};
            unsigned wq[4] = {rq[2 * s][0], rq[2 * s][1],
                              rq[2 * s + 1][0], rq[2 * s + 1][1]};
            bf16x8 dob = *reinterpret_cast<bf16x8*>(wd);
            bf16x8 qb = *reinterpret_cast<bf16x8*>(wq);
            acc_dv[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                pfrag[s], dob, acc_dv[dblk], 0, 0, 0);
            acc_dk[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                dsfrag[s], qb, acc_dk[dblk], 0, 0, 0);
          }
        }
      } else {
        asm volatile("" :: "v"(pfrag[0][0]), "v"(dsfrag[0][0]));
      }
      __syncthreads();
    };

    {
      int t = 0;
      for (; t + 1 < n_iter; t += 2) {
        iter_body(t, rA, rB);
        iter_body(t + 1, rB, rA);
      }
      if (t < n_iter) iter_body(t, rA, rB);
    }

    // dK/dV output layout: col = ln31 = d within block, rows = crow = kv
    // local row. crow covers rows in 4-consecutive groups per (r>>2), but
    // along the D axis the lane's element index is ln31 (stride 1 in d) —
    // the 4-consecutive runs are in the KV direction (r&3), i.e. different
    // rows, so pack via LDS-free transpose is not available; instead swap
    // the roles: for each dblk the lane owns column d=dblk*32+ln31 of 16 kv
    // rows. Pack pairs along r&3 into uint stores per row is not possible
    // (different rows). Keep scalar stores but hoist the bounds check out
    // of the loop (interior workgroups skip all 128 compares).
    const bool all_in = (kvblk0 + 128 <= Tkv);
#pragma unroll
    for (int dblk = 0; dblk < NDBLK; ++dblk)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kvrow = kvblk0 + wid * 32 + crow(r, hi);
        if (all_in || kvrow < Tkv) {
          const long off = kv_base + (long)kvrow * Hkv * D + dblk * 32 + ln31;
          const long dvoff = ((long)b * Tkv + kvrow) * dvp + dvc
              + (long)hkv * D + dblk * 32 + ln31;
          dk[off] = f32_to_bf16(acc_dk[dblk][r]);
          dv[dvoff] = f32_to_bf16(acc_dv[dblk][r]);
        }
      }
  }
};

template <int D>
__global__ __launch_bounds__(256) void dq_kernel(
    const unsigned short* q, const unsigned short* k, const unsigned short* v,
    const unsigned short* dout, const float* lse, const float* delta,
    unsigned short* dq, int B, int Tq, int Tkv, int q_off, int Hq, int Hkv,
    float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  DQKernel<D>::run(q, k, v, dout, lse, delta, dq, B, Tq, Tkv, q_off, Hq, Hkv,
                   scale, smem_raw);
}

template <int D, int ABL = 0>
__global__ __launch_bounds__(256) void dkdv_kernel(
    const unsigned short* q, const unsigned short* k, const unsigned short* v,
    const unsigned short* dout, const float* lse, const float* delta,
    unsigned short* dk, unsigned short* dv, int B, int Tq, int Tkv, int q_off,
    int Hq, int Hkv, float scale, long dvp, long dvc) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  DKDVKernel<D, ABL>::run(q, k, v, dout, lse, delta, dk, dv, B, Tq, Tkv,
                          q_off, Hq, Hkv, scale, dvp, dvc, smem_raw);
}

}  // namespace attnbwd

static std::vector<torch::Tensor> attn_bwd_impl(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor o, torch::Tensor lse, bool causal, long q_offset,
    torch::Tensor dv, long dvp, long dvc) {
  TORCH_CHECK(causal, "attn_bwd: only causal attention is implemented");
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16 && q.is_contiguous());
  const int B = q.size(0), T = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Tkv = k.size(1), Hkv = k.size(2);
  const int q_off = (int)q_offset;
  const float scale = 1.0f / sqrtf((float)D);
  auto stream = at::cuda::getCurrentHIPStream();

  auto delta = torch::empty({B, Hq, T}, q.options().dtype(torch::kFloat32));
  {
    long total_rows = (long)B * T * Hq;
    long waves = total_rows;
    long threads = waves * WAVE_SIZE;
    int block = 256;
    long grid = (threads + block - 1) / block;
    hipLaunchKernelGGL(attnbwd::delta_kernel, dim3((unsigned)grid), dim3(block),
                       0, stream, (const unsigned short*)dout.data_ptr(),
                       (const unsigned short*)o.data_ptr(),
                       delta.data_ptr<float>(), total_rows, T, Hq, D);
    HIP_CHECK_KERNEL();
  }

  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);

  // dq and dkdv are independent given delta: run them on two streams so
  // their partially-filled tail waves overlap (each launches ~1280 blocks
  // over 256 CUs). Raw fork/join events (created once, reused; no timing)
  // keep this hipGraph-capturable.
  static hipStream_t side = nullptr;
  static hipEvent_t ev_fork = nullptr, ev_join = nullptr;
  if (side == nullptr) {
    hipStreamCreateWithFlags(&side, hipStreamNonBlocking);
    hipEventCreateWithFlags(&ev_fork, hipEventDisableTiming);
    hipEventCreateWithFlags(&ev_join, hipEventDisableTiming);
  }
  hipEventRecord(ev_fork, stream);
  hipStreamWaitEvent(side, ev_fork, 0);

  const int n_qblk = (T + 127) / 128;
  const int n_kvblk = (Tkv + 127) / 128;
  if (D == 128) {
    size_t smem_dq = sizeof(typename attnbwd::DQKernel<128>::Smem);
    hipLaunchKernelGGL(attnbwd::dq_kernel<128>, dim3(n_qblk, Hq, B), dim3(256),
                       smem_dq, stream, (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (const unsigned short*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (unsigned short*)dq.data_ptr(), B, T, Tkv, q_off, Hq, Hkv, scale);
    HIP_CHECK_KERNEL();
    size_t smem_kv = sizeof(typename attnbwd::DKDVKernel<128>::Smem);
    hipLaunchKernelGGL(attnbwd::dkdv_kernel<128>, dim3(n_kvblk, Hkv, B),
                       dim3(256), smem_kv, side,
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (const unsigned short*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (unsigned short*)dk.data_ptr(),
                       (unsigned short*)dv.data_ptr(), B, T, Tkv, q_off, Hq, Hkv, scale, dvp, dvc);
    HIP_CHECK_KERNEL();
  } else if (D == 64) {
    size_t smem_dq = sizeof(typename attnbwd::DQKernel<64>::Smem);
    hipLaunchKernelGGL(attnbwd::dq_kernel<64>, dim3(n_qblk, Hq, B), dim3(256),
                       smem_dq, stream, (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (const unsigned short*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (unsigned short*)dq.data_ptr(), B, T, Tkv, q_off, Hq, Hkv, scale);
    HIP_CHECK_KERNEL();
    size_t smem_kv = sizeof(typename attnbwd::DKDVKernel<64>::Smem);
    hipLaunchKernelGGL(attnbwd::dkdv_kernel<64>, dim3(n_kvblk, Hkv, B),
                       dim3(256), smem_kv, side,
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (const unsigned short*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (unsigned short*)dk.data_ptr(),
                       (unsigned short*)dv.data_ptr(), B, T, Tkv, q_off, Hq, Hkv, scale, dvp, dvc);
    HIP_CHECK_KERNEL();
  } else {
    TORCH_CHECK(false, "attn_bwd: head_dim must be 64 or 128");
  }
  hipEventRecord(ev_join, side);
  hipStreamWaitEvent(stream, ev_join, 0);
  return {dq, dk, dv};
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    bool causal, long q_offset) {
  auto dv = torch::empty_like(v);
  const long Hkv = v.size(2), D = v.size(3);
  auto r = attn_bwd_impl(dout, q, k, v, o, lse, causal, q_offset, dv,
                         Hkv * D, 0);
  return {r[0], r[1], dv};
}

// Fused-QKV variant: dv is written directly into its column block of the
// joint gradient buffer dqkv [B, T, Cq+Ck+Cv] (column offset dv_col_off);
// returns {dq, dk} for the RoPE-backward stage to scatter the same way.
std::vector<torch::Tensor> attn_bwd_qkvjoint(torch::Tensor dout,
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
  auto r = attn_bwd_impl(dout, q, k, v, o, lse, causal, q_offset, dqkv,
                         Ctot, dv_col_off);
  return {r[0], r[1]};
}

// Ablation entry: times ONLY the dkdv kernel at the given mode (D=128).
// Modes >0 produce wrong dk/dv; timing only.
std::vector<torch::Tensor> attn_bwd_dkdv_ablate(torch::Tensor dout,
                                                torch::Tensor q,
                                                torch::Tensor k,
                                                torch::Tensor v,
                                                torch::Tensor lse,
                                                torch::Tensor delta,
                                                long mode) {
  const int B = q.size(0), T = q.size(1), Hq = q.size(2), D = q.size(3);
  const int Hkv = k.size(2);
  TORCH_CHECK(D == 128);
  const float scale = 1.0f / sqrtf((float)D);
  auto stream = at::cuda::getCurrentHIPStream();
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  const int n_kvblk = (T + 127) / 128;
  size_t smem_kv = sizeof(typename attnbwd::DKDVKernel<128>::Smem);
  auto launch = [&](auto kfn) {
    hipLaunchKernelGGL(kfn, dim3(n_kvblk, Hkv, B), dim3(256), smem_kv, stream,
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (const unsigned short*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (unsigned short*)dk.data_ptr(),
                       (unsigned short*)dv.data_ptr(), B, T, T, 0, Hq, Hkv,
                       scale, (long)Hkv * 128, 0L);
  };
  switch (mode) {
    case 0: launch(attnbwd::dkdv_kernel<128, 0>); break;
    case 1: launch(attnbwd::dkdv_kernel<128, 1>); break;
    case 2: launch(attnbwd::dkdv_kernel<128, 2>); break;
    case 3: launch(attnbwd::dkdv_kernel<128, 3>); break;
    case 4: launch(attnbwd::dkdv_kernel<128, 4>); break;
    case 5: launch(attnbwd::dkdv_kernel<128, 5>); break;
    default: TORCH_CHECK(false, "mode 0-5");
  }
  HIP_CHECK_KERNEL();
  return {dk, dv};
}
