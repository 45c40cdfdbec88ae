// Full-featured flash-attention BACKWARD (alibi + dropout), separate
// translation unit: the vanilla hot path (flash_attn_fwd.hip) stays at its
// tuned register budget; these variants tolerate spills (guide rule 19:
// co-compiled template variants perturb each other's codegen).
// Flash-attention v2 BACKWARD for gfx950 (CDNA4 MFMA).
//
// Three kernels (recompute variant — deterministic, no atomics):
//   1. fa_bwd_preprocess: delta[b,h,sq] = rowsum(dO * O)
//   2. fa_bwd_dkv: grid over 128-key tiles (4-wave WG, wave = 32 keys);
//      K/V fragments live in registers; per 64-row Q tile (Q, dO, lse,
//      delta staged in LDS) it recomputes P from (S, stored lse), computes
//      dP, dS, and accumulates dV += P^T dO and dK += dS^T Q, using the
//      cvt_pk + permlane32_swap fragment redistribution (T12) to turn the
//      QK^T accumulator layout into mfma A-operands.
//   3. fa_bwd_dq: forward structure (8-wave WG, wave = 32 q rows); K
//      natural + K^T + V natural staged in LDS per 64-key tile; dQ += dS K
//      accumulated in registers.
//
// P is normalized against the stored global lse and delta comes from the
// global (out, dout), so the same kernels serve per-block ring-attention
// backward unchanged.
//
// Fragment-layout cheat sheet (v_mfma_f32_32x32x16_bf16):
//   A[row][k] : lane l holds row = l&31,  k = (l>>5)*8 + j   (j = 0..7)
//   B[k][col] : lane l holds col = l&31,  k = (l>>5)*8 + j
//   D[row][col]: lane l holds col = l&31, row = CROW(r, l>>5), r = 0..15
// t12_pack_frag converts a D-layout f32 pair-block into the bf16 fragment
// whose ROW axis is D's col axis and whose k axis is D's row axis — i.e. it
// transposes the accumulator into an A/B operand.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "attn_common.h"

// ---------------------------------------------------------------------------
// 1. preprocess: delta = rowsum(dO * O)   [b,h,sq] fp32
// ---------------------------------------------------------------------------
template <bool FP16>
__global__ void fa_bwd_preprocess_kernel_x(const short* __restrict__ dO,
                                         const short* __restrict__ O,
                                         float* __restrict__ delta, int b,
                                         int sq, int hq, int D) {
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) +
                   (threadIdx.x / WAVE);
  const int lane = threadIdx.x % WAVE;
  const long rows = (long)b * sq * hq;
  if (row >= rows) return;
  const short* dr = dO + row * D;
  const short* orow = O + row * D;
  float acc = 0.f;
  for (int i = lane * 2; i < D; i += WAVE * 2) {
    acc += AttnElem<FP16>::to_f32(dr[i]) * AttnElem<FP16>::to_f32(orow[i]) +
           AttnElem<FP16>::to_f32(dr[i + 1]) *
               AttnElem<FP16>::to_f32(orow[i + 1]);
  }
  acc = wave_reduce_sum(acc);
  if (lane == 0) {
    const int h = row % hq;
    const long bs = row / hq;
    const long bb = bs / sq, s = bs % sq;
    delta[(bb * hq + h) * sq + s] = acc;
  }
}

// ---------------------------------------------------------------------------
// 2. dK/dV kernel
// ---------------------------------------------------------------------------
template <int D, bool CAUSAL, bool HAS_WINDOW, bool HAS_LENS, bool HAS_EXTRA,
          bool FP16>
__global__ __launch_bounds__(256, 2)
void fa_bwd_dkv_kernel_x(const short* __restrict__ dOut,
                       const short* __restrict__ Q,
                       const short* __restrict__ K,
                       const short* __restrict__ V,
                       const float* __restrict__ LSE,
                       const float* __restrict__ DELTA,
                       short* __restrict__ dK, short* __restrict__ dV,
                       int b_, int sq, int sk, int hq, int hk, float scale,
                       int wl, int wr, const int* __restrict__ q_lens,
                       const int* __restrict__ k_lens,
                       const float* __restrict__ alibi, float p_drop,
                       unsigned long long rng_seed) {
  constexpr int NT = D / 16;
  constexpr int NA = D / 32;
  constexpr int QT = 32;            // q rows per staged tile
  constexpr int KVWG = 128;         // keys per workgroup (4 waves x 32)
  // transposed tiles [D][32]: 64B rows with a ((d>>3)&3)<<4 byte-XOR —
  // staging lanes differ in d by multiples of 8, so the XOR must use bits
  // >=3 of d to spread write banks; reads (b128, 16-lane groups over
  // d = a*32+col) become conflict-free since (d&3, (d>>3)&3) is distinct
  // per lane

  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* q_lds = reinterpret_cast<short*>(smem);            // [QT][D] swz
  short* do_lds = q_lds + QT * D;                           // [QT][D] swz
  short* v_lds = do_lds + QT * D;                           // [KVWG][D] swz
  float* lse_lds = reinterpret_cast<float*>(v_lds + KVWG * D);
  float* del_lds = lse_lds + QT;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;          // 0..3
  const int col = lane & 31;
  const int hi = lane >> 5;

  const int kh = blockIdx.y % hk;
  const int b = blockIdx.y / hk;
  const int gqa = hq / hk;
  // tile<->XCD decorrelation (see fa_fwd_kernel)
  const int kv0wg = (int)((blockIdx.x + blockIdx.y) % gridDim.x) * KVWG;
  const int key_b = kv0wg + wid * 32;          // wave's key block base
  const int mykey = key_b + col;               // lane's key (K frag row)
  const int shift = sk - sq;

  int klimit = sk;
  int qlimit = sq;
  if (HAS_LENS) {
    if (k_lens != nullptr) klimit = min(klimit, k_lens[b]);
    if (q_lens != nullptr) qlimit = min(qlimit, q_lens[b]);
  }
  const bool has_drop = HAS_EXTRA && p_drop > 0.f;
  const float inv_keep = HAS_EXTRA ? 1.f / (1.f - p_drop) : 1.f;
  const unsigned thr24 =
      HAS_EXTRA ? (unsigned)(p_drop * 16777216.f) : 0u;

  // K fragments in registers; V staged once in LDS for the whole workgroup
  bf16x8 kfrag[NT];
  {
    const bool kv_ok = mykey < sk;
    const long base =
        ((long)b * sk + (kv_ok ? mykey : sk - 1)) * hk * D + (long)kh * D;
#pragma unroll
    for (int t = 0; t < NT; ++t) {
      s16x8 kv8 = kv_ok
          ? *reinterpret_cast<const s16x8*>(K + base + t * 16 + hi * 8)
          : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
      for (int j = 0; j < 8; ++j) kfrag[t][j] = kv8[j];
    }
  }
  {
    constexpr int CHUNKS = KVWG * D / 8;
    const long vbase = ((long)b * sk * hk + kh) * D;
    for (int c = tid; c < CHUNKS; c += 256) {
      const int row = c / (D / 8);
      const int d0 = (c % (D / 8)) * 8;
      const int key = kv0wg + row;
      s16x8 vv8 = {0, 0, 0, 0, 0, 0, 0, 0};
      if (key < sk)
        vv8 = *reinterpret_cast<const s16x8*>(V + vbase +
                                              (long)key * hk * D + d0);
      unsigned byte = row * (D * 2) + d0 * 2;
      byte ^= (unsigned)((row & 7) << 4);
      *reinterpret_cast<s16x8*>(reinterpret_cast<char*>(v_lds) + byte) = vv8;
    }
    __syncthreads();
  }

  // per-lane constant address for the tr_b16 B-frag reads
  const int m4 = (lane & 15) >> 2;
  const int dl0 = ((lane >> 4) & 1) * 16 + (lane & 3) * 4;
  const unsigned tr_addr = (unsigned)(size_t)q_lds +
      (unsigned)((hi * 8 + m4) * (D * 2)) +
      (((unsigned)(dl0 * 2)) ^ ((unsigned)(m4 << 4)));

  f32x16 dkacc[NA], dvacc[NA];
#pragma unroll
  for (int a = 0; a < NA; ++a) {
    dkacc[a] = f32x16(0.f);
    dvacc[a] = f32x16(0.f);
  }

  int q_lo = 0;
  if (CAUSAL) q_lo = max(0, kv0wg - shift);
  int q_hi = qlimit;
  if (HAS_WINDOW && wl >= 0)
    q_hi = min(q_hi, kv0wg + KVWG - 1 - shift + wl + 1);
  const int qt0 = q_lo / QT;
  const int qt1 = (max(q_hi, 0) + QT - 1) / QT;

  for (int gh = 0; gh < gqa; ++gh) {
    const int h = kh * gqa + gh;
    const float slope =
        (HAS_EXTRA && alibi != nullptr) ? alibi[h] : 0.f;
    for (int qt = qt0; qt < qt1; ++qt) {
      const int q0 = qt * QT;
      // ---- cooperative staging: natural (swizzled) + transposed tiles ----
      {
        constexpr int CHUNKS = QT * D / 8;
        for (int c = tid; c < CHUNKS; c += 256) {
          const int row = c / (D / 8);
          const int d0 = (c % (D / 8)) * 8;
          const int qrow = q0 + row;
          s16x8 qv = {0, 0, 0, 0, 0, 0, 0, 0};
          s16x8 dv8 = {0, 0, 0, 0, 0, 0, 0, 0};
          if (qrow < sq) {
            const long src = ((long)(b * sq + qrow) * hq + h) * D + d0;
            qv = *reinterpret_cast<const s16x8*>(Q + src);
            dv8 = *reinterpret_cast<const s16x8*>(dOut + src);
          }
          unsigned byte = row * (D * 2) + d0 * 2;
          byte ^= (unsigned)((row & 7) << 4);
          *reinterpret_cast<s16x8*>(reinterpret_cast<char*>(q_lds) + byte) =
              qv;
          *reinterpret_cast<s16x8*>(reinterpret_cast<char*>(do_lds) + byte) =
              dv8;
        }
        for (int r = tid; r < QT; r += 256) {
          const int qrow = q0 + r;
          const long idx = ((long)b * hq + h) * sq + min(qrow, sq - 1);
          lse_lds[r] = (qrow < sq) ? LSE[idx] : INFINITY;
          del_lds[r] = (qrow < sq) ? DELTA[idx] : 0.f;
        }
      }
      __syncthreads();

      // S = Q K^T and dP = dO V^T, both D[q=CROW][key=col]
      f32x16 s = f32x16(0.f);
      f32x16 dp = f32x16(0.f);
#pragma unroll
      for (int t = 0; t < NT; ++t) {
        const int row = col;
        unsigned byte = row * (D * 2) + (t * 16 + hi * 8) * 2;
        byte ^= (unsigned)((row & 7) << 4);
        bf16x8 qf = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(q_lds) + byte);
        bf16x8 df = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(do_lds) + byte);
        const int vrow = wid * 32 + col;
        unsigned vbyte = vrow * (D * 2) + (t * 16 + hi * 8) * 2;
        vbyte ^= (unsigned)((vrow & 7) << 4);
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(v_lds) + vbyte);
        s = AttnElem<FP16>::mfma(qf, kfrag[t], s);
        dp = AttnElem<FP16>::mfma(df, vf, dp);
      }
      // P = exp(S*scale - lse); dS = P*(dP - delta)*scale
      // (s reused as P, dp reused as dS — register budget)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = q0 + CROW(r, hi);
        const float lse_q = lse_lds[CROW(r, hi)];
        const float del_q = del_lds[CROW(r, hi)];
        bool valid =
            (qrow < qlimit) && (mykey < klimit) && isfinite(lse_q);
        if (CAUSAL) valid &= (mykey <= qrow + shift);
        if (HAS_WINDOW && wl >= 0) valid &= (mykey >= qrow + shift - wl);
        if (HAS_WINDOW && wr >= 0 && !CAUSAL)
          valid &= (mykey <= qrow + shift + wr);
        float sv = s[r] * scale;
        if (HAS_EXTRA && alibi != nullptr)
          sv -= slope * fabsf((float)(qrow + shift - mykey));
        const float p = valid ? __expf(sv - lse_q) : 0.f;
        if (HAS_EXTRA && has_drop) {
          const unsigned long long idx =
              (((unsigned long long)(b * hq + h) * sq + qrow)) *
                  (unsigned long long)sk + mykey;
          const float keep =
              attn_dropout_keep(rng_seed, idx, thr24) ? inv_keep : 0.f;
          // dV uses the dropped P; dS = P * (masked dP - delta)
          s[r] = p * keep;
          dp[r] = p * (dp[r] * keep - del_q) * scale;
        } else {
          s[r] = p;
          dp[r] = p * (dp[r] - del_q) * scale;
        }
      }
      // B-frags via ds_read_b64_tr_b16 from the natural swizzled images
      // (see flash_attn_bwd.hip DKV_TR_STEP for the address derivation)
#define DKVX_TR_STEP(tp_, a_)                                              \
      {                                                                    \
        attn_u32x2 ql_, qh_, dl_, dh_;                                     \
        asm volatile(                                                      \
            "ds_read_b64_tr_b16 %0, %4 offset:%c5\n\t"                   \
            "ds_read_b64_tr_b16 %1, %4 offset:%c6\n\t"                   \
            "ds_read_b64_tr_b16 %2, %4 offset:%c7\n\t"                   \
            "ds_read_b64_tr_b16 %3, %4 offset:%c8\n\t"                   \
            "s_waitcnt lgkmcnt(0)"                                         \
            : "=v"(ql_), "=v"(qh_), "=v"(dl_), "=v"(dh_)                   \
            : "v"(tr_addr),                                                \
              "i"((tp_) * 16 * D * 2 + (a_) * 64),                         \
              "i"((tp_) * 16 * D * 2 + 4 * D * 2 + ((a_) ^ 1) * 64),       \
              "i"(QT * D * 2 + (tp_) * 16 * D * 2 + (a_) * 64),            \
              "i"(QT * D * 2 + (tp_) * 16 * D * 2 + 4 * D * 2 +            \
                  ((a_) ^ 1) * 64));                                       \
        attn_u32x4 uq_ = {ql_.x, ql_.y, qh_.x, qh_.y};                     \
        attn_u32x4 ud_ = {dl_.x, dl_.y, dh_.x, dh_.y};                     \
        bf16x8 qbf = __builtin_bit_cast(bf16x8, uq_);                      \
        bf16x8 dob = __builtin_bit_cast(bf16x8, ud_);                      \
        dvacc[a_] = AttnElem<FP16>::mfma(pb, dob, dvacc[a_]);              \
        dkacc[a_] = AttnElem<FP16>::mfma(dsb, qbf, dkacc[a_]);             \
      }
#pragma unroll
      for (int tp = 0; tp < 2; ++tp) {
        unsigned pfr[4], dsfr[4];
        t12_pack_frag<AttnElem<FP16>>(s, tp, pfr);
        t12_pack_frag<AttnElem<FP16>>(dp, tp, dsfr);
        bf16x8 pb = *reinterpret_cast<const bf16x8*>(pfr);
        bf16x8 dsb = *reinterpret_cast<const bf16x8*>(dsfr);
        if (tp == 0) {
          DKVX_TR_STEP(0, 0);
          DKVX_TR_STEP(0, 1);
          if constexpr (NA > 2) {
            DKVX_TR_STEP(0, 2);
            DKVX_TR_STEP(0, 3);
          }
        } else {
          DKVX_TR_STEP(1, 0);
          DKVX_TR_STEP(1, 1);
          if constexpr (NA > 2) {
            DKVX_TR_STEP(1, 2);
            DKVX_TR_STEP(1, 3);
          }
        }
      }
#undef DKVX_TR_STEP
      __syncthreads();
    }
  }

  // ---- epilogue: D[key = key_b + CROW(r,hi)][d = a*32 + col] ------------
#pragma unroll
  for (int a = 0; a < NA; ++a) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key = key_b + CROW(r, hi);
      const int d = a * 32 + col;
      if (key < sk) {
        const long base = ((long)b * sk + key) * hk * D + (long)kh * D;
        dK[base + d] = AttnElem<FP16>::from_f32(dkacc[a][r]);
        dV[base + d] = AttnElem<FP16>::from_f32(dvacc[a][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 3. dQ kernel (forward structure)
// ---------------------------------------------------------------------------
template <int D, bool CAUSAL, bool HAS_WINDOW, bool HAS_LENS, bool HAS_EXTRA,
          bool FP16>
__global__ __launch_bounds__(512, 2)
void fa_bwd_dq_kernel_x(const short* __restrict__ dOut,
                      const short* __restrict__ Q,
                      const short* __restrict__ K,
                      const short* __restrict__ V,
                      const float* __restrict__ LSE,
                      const float* __restrict__ DELTA,
                      short* __restrict__ dQ, int b_, int sq, int sk, int hq,
                      int hk, float scale, int wl, int wr,
                      const int* __restrict__ q_lens,
                      const int* __restrict__ k_lens,
                      const float* __restrict__ alibi, float p_drop,
                      unsigned long long rng_seed) {
  constexpr int NT = D / 16;
  constexpr int NA = D / 32;
  constexpr int KVB = 64;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* k_lds = reinterpret_cast<short*>(smem);     // [64][D] swz
  short* kt_lds = k_lds + KVB * D;                   // [D][64] swz
  short* v_lds = kt_lds + D * KVB;                   // [64][D] swz

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;
  const int hi = lane >> 5;

  const int h = blockIdx.y % hq;
  const int b = blockIdx.y / hq;
  const int kh = h / (hq / hk);
  // tile<->XCD decorrelation (see fa_fwd_kernel)
  const int q0wg = (int)((blockIdx.x + blockIdx.y) % gridDim.x) * 256;
  const int q0 = q0wg + wid * 32;
  const int qrow = q0 + col;
  const int shift = sk - sq;

  int klimit = sk;
  int qlimit = sq;
  if (HAS_LENS) {
    if (k_lens != nullptr) klimit = min(klimit, k_lens[b]);
    if (q_lens != nullptr) qlimit = min(qlimit, q_lens[b]);
  }

  bf16x8 qfrag[NT], dofrag[NT];
  {
    const long qbase = ((long)(b * sq + min(qrow, sq - 1)) * hq + h) * D;
    const bool qvalid = (qrow < sq);
#pragma unroll
    for (int t = 0; t < NT; ++t) {
      s16x8 v8 = qvalid
          ? *reinterpret_cast<const s16x8*>(Q + qbase + t * 16 + hi * 8)
          : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
      s16x8 d8 = qvalid
          ? *reinterpret_cast<const s16x8*>(dOut + qbase + t * 16 + hi * 8)
          : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        qfrag[t][j] = v8[j];
        dofrag[t][j] = d8[j];
      }
    }
  }
  const bool row_ok = (qrow < qlimit);
  const bool has_drop = HAS_EXTRA && p_drop > 0.f;
  const float inv_keep = HAS_EXTRA ? 1.f / (1.f - p_drop) : 1.f;
  const unsigned thr24 =
      HAS_EXTRA ? (unsigned)(p_drop * 16777216.f) : 0u;
  const float slope =
      (HAS_EXTRA && alibi != nullptr) ? alibi[h] : 0.f;
  const float lse_q =
      row_ok ? LSE[((long)b * hq + h) * sq + qrow] : INFINITY;
  const float del_q =
      row_ok ? DELTA[((long)b * hq + h) * sq + qrow] : 0.f;

  f32x16 dqacc[NA];
#pragma unroll
  for (int a = 0; a < NA; ++a) dqacc[a] = f32x16(0.f);

  int kv_hi_key = klimit;
  if (CAUSAL) kv_hi_key = min(kv_hi_key, q0wg + 255 + shift + 1);
  int kv_lo_key = 0;
  if (HAS_WINDOW && wl >= 0) kv_lo_key = max(0, q0wg + shift - wl);
  const int t0 = kv_lo_key / KVB;
  const int t1 = (max(kv_hi_key, 0) + KVB - 1) / KVB;

  const long kbase = ((long)b * sk * hk + kh) * D;
  for (int tile = t0; tile < t1; ++tile) {
    const int kv0 = tile * KVB;
    {
      constexpr int CHUNKS = KVB * D / 8;
      for (int c = tid; c < CHUNKS; c += 512) {
        const int row = c / (D / 8);
        const int d0 = (c % (D / 8)) * 8;
        const int key = kv0 + row;
        s16x8 kv8 = {0, 0, 0, 0, 0, 0, 0, 0};
        s16x8 vv8 = {0, 0, 0, 0, 0, 0, 0, 0};
        if (key < sk) {
          const long src = kbase + (long)key * hk * D + d0;
          kv8 = *reinterpret_cast<const s16x8*>(K + src);
          vv8 = *reinterpret_cast<const s16x8*>(V + src);
        }
        unsigned byte = row * (D * 2) + d0 * 2;
        byte ^= (unsigned)((row & 7) << 4);
        *reinterpret_cast<s16x8*>(reinterpret_cast<char*>(k_lds) + byte) =
            kv8;
        *reinterpret_cast<s16x8*>(reinterpret_cast<char*>(v_lds) + byte) =
            vv8;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          unsigned tbyte = (unsigned)(d0 + j) * (KVB * 2) + row * 2;
          tbyte ^= (unsigned)((((d0 + j) >> 3) & 7) << 4);
          *reinterpret_cast<short*>(reinterpret_cast<char*>(kt_lds) +
                                    tbyte) = kv8[j];
        }
      }
    }
    __syncthreads();

    bool wave_active = true;
    if (CAUSAL && kv0 > q0 + 31 + shift) wave_active = false;
    if (HAS_WINDOW && wl >= 0 && kv0 + KVB <= q0 + shift - wl)
      wave_active = false;

    if (wave_active) {
      // swapped S and dP^T: D[key=CROW][q=col]
      f32x16 s[2], dp[2];
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
        s[kb] = f32x16(0.f);
        dp[kb] = f32x16(0.f);
      }
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
        for (int t = 0; t < NT; ++t) {
          const int row = kb * 32 + col;
          unsigned byte = row * (D * 2) + (t * 16 + hi * 8) * 2;
          byte ^= (unsigned)((row & 7) << 4);
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(k_lds) + byte);
          bf16x8 vf = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(v_lds) + byte);
          s[kb] = AttnElem<FP16>::mfma(kf, qfrag[t], s[kb]);
          dp[kb] = AttnElem<FP16>::mfma(vf, dofrag[t], dp[kb]);
        }
      }
      // reuse dp as dS (register budget)
      const bool lane_ok = row_ok && isfinite(lse_q);
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int key = kv0 + kb * 32 + CROW(r, hi);
          bool valid = lane_ok && (key < klimit);
          if (CAUSAL) valid &= (key <= qrow + shift);
          if (HAS_WINDOW && wl >= 0) valid &= (key >= qrow + shift - wl);
          if (HAS_WINDOW && wr >= 0 && !CAUSAL)
            valid &= (key <= qrow + shift + wr);
          float sv = s[kb][r] * scale;
          if (HAS_EXTRA && alibi != nullptr)
            sv -= slope * fabsf((float)(qrow + shift - key));
          const float p = valid ? __expf(sv - lse_q) : 0.f;
          float dpv = dp[kb][r];
          if (HAS_EXTRA && has_drop) {
            const unsigned long long idx =
                (((unsigned long long)(b * hq + h) * sq + qrow)) *
                    (unsigned long long)sk + key;
            dpv *= attn_dropout_keep(rng_seed, idx, thr24) ? inv_keep : 0.f;
          }
          dp[kb][r] = p * (dpv - del_q) * scale;
        }
      }
      // dQ[q][d] += dS[q][key] K[key][d]
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
        for (int tp = 0; tp < 2; ++tp) {
          unsigned dsfr[4];
          t12_pack_frag<AttnElem<FP16>>(dp[kb], tp, dsfr);
          bf16x8 dsb = *reinterpret_cast<const bf16x8*>(dsfr);
#pragma unroll
          for (int a = 0; a < NA; ++a) {
            const int d = a * 32 + col;
            unsigned byte = (unsigned)d * (KVB * 2) +
                            (kb * 32 + 16 * tp + hi * 8) * 2;
            byte ^= (unsigned)(((d >> 3) & 7) << 4);
            bf16x8 ktb = *reinterpret_cast<const bf16x8*>(
                reinterpret_cast<const char*>(kt_lds) + byte);
            dqacc[a] = AttnElem<FP16>::mfma(dsb, ktb, dqacc[a]);
          }
        }
      }
    }
    __syncthreads();
  }

  // epilogue: D[q = q0 + CROW(r,hi)][d = a*32 + col]
#pragma unroll
  for (int a = 0; a < NA; ++a) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int q_r = q0 + CROW(r, hi);
      const int d = a * 32 + col;
      if (q_r < sq) {
        const long obase = ((long)(b * sq + q_r) * hq + h) * D;
        dQ[obase + d] = (q_r < qlimit)
            ? AttnElem<FP16>::from_f32(dqacc[a][r]) : (short)0;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

template <int D, bool FP16>
static void launch_fa_bwd_x(const torch::Tensor& dout, const torch::Tensor& q,
                          const torch::Tensor& k, const torch::Tensor& v,
                          const torch::Tensor& lse,
                          const torch::Tensor& delta, torch::Tensor& dq,
                          torch::Tensor& dk, torch::Tensor& dv, float scale,
                          bool causal, int wl, int wr, const int* qlp,
                          const int* klp, const float* alp, float p_drop,
                          unsigned long long rng_seed,
                          hipStream_t stream) {
  const int b = q.size(0), sq = q.size(1), hq = q.size(2);
  const int sk = k.size(1), hk = k.size(2);
  const bool has_window = (wl >= 0 || wr >= 0);
  const bool has_lens = qlp != nullptr || klp != nullptr;

#define ARGS_DKV                                                             \
  (const short*)dout.data_ptr(), (const short*)q.data_ptr(),                 \
      (const short*)k.data_ptr(), (const short*)v.data_ptr(),                \
      lse.data_ptr<float>(), delta.data_ptr<float>(),                        \
      (short*)dk.data_ptr(), (short*)dv.data_ptr(), b, sq, sk, hq, hk,       \
      scale, wl, wr, qlp, klp, alp, p_drop, rng_seed
#define ARGS_DQ                                                              \
  (const short*)dout.data_ptr(), (const short*)q.data_ptr(),                 \
      (const short*)k.data_ptr(), (const short*)v.data_ptr(),                \
      lse.data_ptr<float>(), delta.data_ptr<float>(),                        \
      (short*)dq.data_ptr(), b, sq, sk, hq, hk, scale, wl, wr, qlp, klp,    \
      alp, p_drop, rng_seed

  dim3 gkv((sk + 127) / 128, b * hk), bkv(256);
  const int lds_kv = (2 * 32 * D + 128 * D) * 2 + 2 * 32 * 4;
  dim3 gq((sq + 255) / 256, b * hq), bq(512);
  const int lds_q = 3 * 64 * D * 2;

#define DISPATCH_CASE(C, W, L, DR)                                          \
  if (causal == C && has_window == W && has_lens == L && has_drop == DR) {   \
    hipLaunchKernelGGL((fa_bwd_dkv_kernel_x<D, C, W, L, DR, FP16>), gkv,    \
                       bkv, lds_kv, stream, ARGS_DKV);                       \
    hipLaunchKernelGGL((fa_bwd_dq_kernel_x<D, C, W, L, DR, FP16>), gq, bq,   \
                       lds_q, stream, ARGS_DQ);                              \
    return;                                                                  \
  }
  const bool has_drop = true;  // extra TU: alibi/dropout active
  DISPATCH_CASE(false, false, false, true)
  DISPATCH_CASE(false, false, true, true)
  DISPATCH_CASE(false, true, false, true)
  DISPATCH_CASE(false, true, true, true)
  DISPATCH_CASE(true, false, false, true)
  DISPATCH_CASE(true, false, true, true)
  DISPATCH_CASE(true, true, false, true)
  DISPATCH_CASE(true, true, true, true)
#undef DISPATCH_CASE
#undef ARGS_DKV
#undef ARGS_DQ
}

std::vector<torch::Tensor> fa_backward_extra(torch::Tensor dout, torch::Tensor q,
                                       torch::Tensor k, torch::Tensor v,
                                       torch::Tensor out, torch::Tensor lse,
                                       double softmax_scale, bool causal,
                                       long wl, long wr, torch::Tensor q_lens,
                                       torch::Tensor k_lens,
                                       torch::Tensor alibi_slopes,
                                       double p_drop, long rng_seed) {
  const bool fp16 = q.scalar_type() == torch::kHalf;
  TORCH_CHECK(q.is_cuda() && (q.scalar_type() == torch::kBFloat16 || fp16),
              "fa_backward: bf16 only");
  TORCH_CHECK(dout.is_contiguous() && q.is_contiguous() &&
              k.is_contiguous() && v.is_contiguous());
  const int b = q.size(0), sq = q.size(1), hq = q.size(2), D = q.size(3);
  TORCH_CHECK(D == 64 || D == 128);
  auto stream = at::hip::getCurrentHIPStream();
  auto lse_c = lse.contiguous();
  auto delta = torch::empty_like(lse_c);
  {
    const long rows = (long)b * sq * hq;
    const long grid = (rows + 3) / 4;
    if (fp16)
      hipLaunchKernelGGL(fa_bwd_preprocess_kernel_x<true>,
                         dim3((unsigned)grid), dim3(256), 0, stream,
                         (const short*)dout.data_ptr(),
                         (const short*)out.data_ptr(),
                         delta.data_ptr<float>(), b, sq, hq, D);
    else
      hipLaunchKernelGGL(fa_bwd_preprocess_kernel_x<false>,
                         dim3((unsigned)grid), dim3(256), 0, stream,
                         (const short*)dout.data_ptr(),
                         (const short*)out.data_ptr(),
                         delta.data_ptr<float>(), b, sq, hq, D);
  }
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto ql = q_lens.numel() ? q_lens.to(q.device(), torch::kInt32) : q_lens;
  auto kl = k_lens.numel() ? k_lens.to(q.device(), torch::kInt32) : k_lens;
  const int* qlp = ql.numel() ? ql.data_ptr<int>() : nullptr;
  const int* klp = kl.numel() ? kl.data_ptr<int>() : nullptr;
  auto al = alibi_slopes.numel()
                ? alibi_slopes.to(q.device(), torch::kFloat32).contiguous()
                : alibi_slopes;
  const float* alp = al.numel() ? al.data_ptr<float>() : nullptr;
  if (D == 128) {
    if (fp16)
      launch_fa_bwd_x<128, true>(dout, q, k, v, lse_c, delta, dq, dk, dv,
                                 (float)softmax_scale, causal, (int)wl,
                                 (int)wr, qlp, klp, alp, (float)p_drop,
                                 (unsigned long long)rng_seed, stream);
    else
      launch_fa_bwd_x<128, false>(dout, q, k, v, lse_c, delta, dq, dk, dv,
                                  (float)softmax_scale, causal, (int)wl,
                                  (int)wr, qlp, klp, alp, (float)p_drop,
                                  (unsigned long long)rng_seed, stream);
  } else {
    if (fp16)
      launch_fa_bwd_x<64, true>(dout, q, k, v, lse_c, delta, dq, dk, dv,
                                (float)softmax_scale, causal, (int)wl,
                                (int)wr, qlp, klp, alp, (float)p_drop,
                                (unsigned long long)rng_seed, stream);
    else
      launch_fa_bwd_x<64, false>(dout, q, k, v, lse_c, delta, dq, dk, dv,
                                 (float)softmax_scale, causal, (int)wl,
                                 (int)wr, qlp, klp, alp, (float)p_drop,
                                 (unsigned long long)rng_seed, stream);
  }
  HIP_CHECK_LAST();
  return {dq, dk, dv};
}
