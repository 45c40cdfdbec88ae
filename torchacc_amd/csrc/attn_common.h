// Shared helpers for the CDNA4 attention kernels.
#pragma once
#include "common.h"

#define LOG2E_F 1.4426950408889634f

// D-matrix row for accumulator register r on lane-half hi (32x32 mfma):
// row = (r&3) + 8*(r>>2) + 4*hi
#define CROW(r, hi) (((r) & 3) + 8 * ((r) >> 2) + 4 * (hi))

DEVINLINE unsigned attn_cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
               : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

typedef _Float16 attn_f16x8 __attribute__((ext_vector_type(8)));
typedef __fp16 attn_fp16x2 __attribute__((ext_vector_type(2)));

// Element traits: the attention kernels are templated on FP16 so the SAME
// MFMA/LDS structure serves bf16 (v_mfma_..._bf16, v_cvt_pk_bf16_f32) and
// fp16 (v_mfma_..._f16, v_cvt_pkrtz_f16_f32). Storage stays short-based
// (16-bit) everywhere — only the convert/mfma ops differ (reference
// accepts fp16 and bf16, ops/flash_attn.py:324-325).
template <bool FP16>
struct AttnElem;

template <>
struct AttnElem<false> {  // bf16
  static DEVINLINE float to_f32(short u) { return bf16_to_f32(u); }
  static DEVINLINE short from_f32(float f) { return f32_to_bf16(f); }
  static DEVINLINE unsigned cvt_pk(float lo, float hi) {
    return attn_cvt_pk_bf16(lo, hi);
  }
  template <typename A, typename ACC>
  static DEVINLINE ACC mfma(A a, A b, ACC acc) {
    return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  }
};

template <>
struct AttnElem<true> {  // fp16
  static DEVINLINE float to_f32(short u) {
    __half h = __builtin_bit_cast(__half, u);
    return __half2float(h);
  }
  static DEVINLINE short from_f32(float f) {
    __half h = __float2half(f);
    return __builtin_bit_cast(short, h);
  }
  static DEVINLINE unsigned cvt_pk(float lo, float hi) {
    attn_fp16x2 h = __builtin_amdgcn_cvt_pkrtz(lo, hi);
    return __builtin_bit_cast(unsigned, h);
  }
  template <typename A, typename ACC>
  static DEVINLINE ACC mfma(A a, A b, ACC acc) {
    attn_f16x8 af = __builtin_bit_cast(attn_f16x8, a);
    attn_f16x8 bf = __builtin_bit_cast(attn_f16x8, b);
    return __builtin_amdgcn_mfma_f32_32x32x16_f16(af, bf, acc, 0, 0, 0);
  }
};

// T12 redistribution: given a 32-wide f32 accumulator pair layout
// D[crow(r,hi)][col=lane&31], produce the 16-bit x8 B/A-fragment for mfma
// step tp (16 rows [16*tp, 16*tp+16)): lane half hi receives rows
// 16tp+8hi+j. Returns 4 packed u32 (8 elems) in frag[0..3].
template <typename ET, typename V16>
DEVINLINE void t12_pack_frag(const V16& p16 /*16 regs*/, int tp,
                             unsigned* frag) {
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    const int r = 2 * u + 8 * tp;
    unsigned va = ET::cvt_pk(p16[r], p16[r + 1]);
    unsigned vb = ET::cvt_pk(p16[r + 4], p16[r + 5]);
    auto sw = __builtin_amdgcn_permlane32_swap(va, vb, false, false);
    frag[u] = sw[0];
    frag[u + 2] = sw[1];
  }
}


typedef unsigned attn_u32x2 __attribute__((ext_vector_type(2)));
typedef unsigned attn_u32x4 __attribute__((ext_vector_type(4)));

// gfx950 ds_read_b64_tr_b16: each lane fetches 4 contiguous bf16 (8 B) at
// its own LDS byte address; the hardware transposes the 16-lane group's
// [16 lane][4 elem] fetch matrix to [4][16], delivering column (lane&15).
// Measured semantics (csrc/tools/tr16_probe.hip): out[m][j] = the element
// fetched by lane ((m+16j)>>2) at slot ((m+16j)&3) of the same group —
// so with lane addresses  addr(l) = &X[q0 + ((l&15)>>2)][d0 + (l&3)*4]
// the read delivers frag slots j=0..3 of the B-fragment column
// X[q0+j][d0 + (l&15)] (row-position swizzles are free: addresses are
// per-lane). Two reads (q0, q0+4) build a full bf16x8 B-frag.
DEVINLINE attn_u32x2 attn_tr16_pair(unsigned a0, unsigned a1,
                                    attn_u32x2& hi) {
  attn_u32x2 lo;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=v"(lo), "=v"(hi)
      : "v"(a0), "v"(a1));
  return lo;
}

// counter-based dropout RNG (splitmix64): deterministic keep-decision per
// (seed, flat attention index) — the backward kernels regenerate the same
// mask the forward applied (reference: FA2 philox rng_state).
DEVINLINE bool attn_dropout_keep(unsigned long long seed,
                                 unsigned long long idx,
                                 unsigned threshold24) {
  unsigned long long x = seed + idx * 0x9E3779B97F4A7C15ull;
  x ^= x >> 30; x *= 0xBF58476D1CE4E5B9ull;
  x ^= x >> 27; x *= 0x94D049BB133111EBull;
  x ^= x >> 31;
  return ((unsigned)x & 0xFFFFFFu) >= threshold24;
}

// host-side dtype dispatch: runs BODY with constexpr kFP16 bound
#define FP16_SWITCH(COND, ...)                         \
  do {                                                 \
    if (COND) {                                        \
      constexpr bool kFP16 = true;                     \
      __VA_ARGS__;                                     \
    } else {                                           \
      constexpr bool kFP16 = false;                    \
      __VA_ARGS__;                                     \
    }                                                  \
  } while (0)
