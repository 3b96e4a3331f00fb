// Shared fragment-map helpers for the K1 attention kernels (fwd + bwd).
// All maps were verified on gfx950 by the probes in attention_fwd.hip
// (mfma_probe_32x32x16 / tr16_probe).
#pragma once
#include <type_traits>

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(2))) unsigned int uintx2;

namespace attnc {

// ds_read_b64_tr_b16 (semantics verified by tr16_probe on gfx950): with the
// [panel][row][16-col] image and per-lane address
//   base + (panel*PR*16 + (k0 + (lambda>>2))*16 + 4*(lambda&3)) * 2B,
// lambda = lane&15, the read returns to lane l element j = X[k0+j][d0+(l&15)]
// — the 32x32x16 MFMA A/B fragment k-run (4 of 8 elements; reads at k0 and
// k0+4 complete it). Conflict-free: 16 lanes hit 16 distinct even dwords.
__device__ __forceinline__ uintx2 tr_read_b64(unsigned addr_bytes) {
  uintx2 r;
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(r) : "v"(addr_bytes));
  return r;
}

// Immediate-offset variant: ONE base VGPR per kernel + compile-time
// offsets keep the per-tile transpose reads free of address VALU (the
// base is tile-invariant with a single LDS buffer).
template <int OFF>
__device__ __forceinline__ uintx2 tr_read_b64_off(unsigned base_bytes) {
  static_assert(OFF >= 0 && OFF < 65536, "ds offset is 16-bit");
  uintx2 r;
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:%c2"
               : "=v"(r) : "v"(base_bytes), "i"(OFF));
  return r;
}

// compile-time-unrolled loop whose induction variable is a constant
// expression (usable as a template argument, e.g. for ds offset:N).
template <int I, int N, typename F>
__device__ __forceinline__ void static_for_impl(F&& f) {
  if constexpr (I < N) {
    f(std::integral_constant<int, I>{});
    static_for_impl<I + 1, N>(f);
  }
}
template <int N, typename F>
__device__ __forceinline__ void static_for(F&& f) {
  static_for_impl<0, N>(f);
}

__device__ __forceinline__ unsigned lds_addr(const void* p) {
  return (unsigned)(unsigned long long)p;  // LDS aperture: low 32 = offset
}

// Row-major tile XOR swizzle (guide T2/G4): b128 reads of rows that differ
// within a 16-lane group spread over distinct 16B slots.
__device__ __forceinline__ int swz(int row, int col) {
  return col ^ (((row & 3) | (((row >> 4) & 1) << 2)) << 3);
}

// C/D register map of v_mfma_f32_32x32x16_bf16 (guide §3):
//   col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
__device__ __forceinline__ int crow(int reg, int hi) {
  return (reg & 3) + 8 * (reg >> 2) + 4 * hi;
}

// One v_cvt_pk_bf16_f32 per f32 pair (hipcc fuses adjacent __bf16 casts;
// the manual round-to-nearest-even bit-twiddle of common.h costs ~6 VALU
// per element and dominated the v2 kernels' issue budget).
__device__ __forceinline__ unsigned int cvt_pk_bf16(float lo, float hi) {
  union { __bf16 h[2]; unsigned int u; } c;
  c.h[0] = (__bf16)lo;
  c.h[1] = (__bf16)hi;
  return c.u;
}

// Re-layout f32 values from the MFMA C layout (col = lane&31 = j fixed,
// rows i = crow(reg,hi)) into bf16 A/B fragments with per-lane element
// k = hi*8+m (+16 per frag): in-register transpose via cvt_pk +
// permlane32_swap (guide T12). nfrag = source_rows/16.
__device__ __forceinline__ void c_layout_to_frags(const float* p, bf16x8* frag,
                                                  int nfrag) {
  for (int s = 0; s < nfrag; ++s) {
    unsigned int w[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      w[i] = cvt_pk_bf16(p[8 * s + 2 * i], p[8 * s + 2 * i + 1]);
    auto r02 = __builtin_amdgcn_permlane32_swap(w[0], w[2], false, false);
    auto r13 = __builtin_amdgcn_permlane32_swap(w[1], w[3], false, false);
    unsigned int fw[4] = {(unsigned int)r02[0], (unsigned int)r13[0],
                          (unsigned int)r02[1], (unsigned int)r13[1]};
    frag[s] = *reinterpret_cast<bf16x8*>(fw);
  }
}

// Same re-layout, but straight from a C-accumulator vector with static
// indexing (keeps everything in registers — no intermediate float[16]
// array, which costs 16 VGPRs of live range per tensor in tight kernels).
__device__ __forceinline__ void c16_to_frags(const floatx16& c, bf16x8* frag) {
#pragma unroll
  for (int s = 0; s < 2; ++s) {
    unsigned int w[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
      w[i] = cvt_pk_bf16(c[8 * s + 2 * i], c[8 * s + 2 * i + 1]);
    auto r02 = __builtin_amdgcn_permlane32_swap(w[0], w[2], false, false);
    auto r13 = __builtin_amdgcn_permlane32_swap(w[1], w[3], false, false);
    unsigned int fw[4] = {(unsigned int)r02[0], (unsigned int)r13[0],
                          (unsigned int)r02[1], (unsigned int)r13[1]};
    frag[s] = *reinterpret_cast<bf16x8*>(fw);
  }
}

}  // namespace attnc
