// Fused VARLEN (packed-sequence) flash-attention for gfx950 (CDNA4 MFMA).
//
// One kernel launch for a whole packed batch (total_tokens rows, any number
// of sequences) instead of the per-sequence host loop: the reference exposes
// this as flash_attn_varlen_(qkv)packed_xla over cu_seqlens
// (torchacc/ops/flash_attn.py:24-210). The MI355X-native trick that makes
// the fused path cheap: with q and k packed IDENTICALLY, the "same
// sequence" predicate for a (q,row key) pair collapses to a per-Q-ROW
// interval test  key in [row_start, row_end)  — so each lane loads its own
// row's int2 bounds ONCE and no per-key sequence lookup exists anywhere in
// the inner loops. Workgroup/wave tile ranges come from the bounds of the
// first/last row the wave owns (bounds are monotone over packed rows).
//
// Kernels are structural clones of flash_attn_fwd.hip / flash_attn_bwd.hip
// (same MFMA/LDS layout decisions — see those files for the derivations)
// with {causal(no shift), seq-interval} masking and b=1 indexing. Covers
// GQA and D in {64,128}, bf16; window/alibi/dropout fall back to the
// per-sequence loop in ops/flash_attn.py.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "attn_common.h"

typedef float f32x16_ __attribute__((ext_vector_type(16)));

// ---------------------------------------------------------------------------
// 1. forward
// ---------------------------------------------------------------------------
template <int D, bool CAUSAL, bool FP16>
__global__ __launch_bounds__(512, 2)
void fa_vl_fwd_kernel(const short* __restrict__ Q,
                      const short* __restrict__ K,
                      const short* __restrict__ V, short* __restrict__ O,
                      float* __restrict__ LSE,
                      const int2* __restrict__ BND, int total, int hq,
                      int hk, float scale) {
  constexpr int NT = D / 16;
  constexpr int NA = D / 32;
  constexpr int KVB = 64;
  constexpr int KROW_BYTES = D * 2;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* k_lds = reinterpret_cast<short*>(smem);
  short* vt_lds = reinterpret_cast<short*>(smem) + 2 * KVB * D;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;
  const int hi = lane >> 5;

  const int h = blockIdx.y;
  const int kh = h / (hq / hk);
  const int q0wg = (int)((blockIdx.x + blockIdx.y) % gridDim.x) * 256;
  const int q0 = q0wg + wid * 32;
  const int qrow = q0 + col;

  // per-lane sequence interval of this lane's q row
  int2 myb = BND[min(qrow, total - 1)];
  if (qrow >= total) myb = int2{0, 0};
  // wave scalars: bounds of the wave's first/last row (monotone)
  const int2 wb0 = BND[min(q0, total - 1)];
  const int2 wb31 = BND[min(q0 + 31, total - 1)];

  bf16x8 qfrag[NT];
  {
    const long qbase = ((long)min(qrow, total - 1) * hq + h) * D;
    const bool qvalid = (qrow < total);
#pragma unroll
    for (int t = 0; t < NT; ++t) {
      s16x8 v = qvalid
          ? *reinterpret_cast<const s16x8*>(Q + qbase + t * 16 + hi * 8)
          : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
      for (int j = 0; j < 8; ++j) qfrag[t][j] = v[j];
    }
  }

  // ---- KV tile range (workgroup): first row's seq start .. last row's
  // seq end (or the causal diagonal) -------------------------------------
  const int qend = min(q0wg + 255, total - 1);
  const int kv_lo_key = BND[min(q0wg, total - 1)].x;
  const int kv_hi_key = CAUSAL ? (qend + 1) : BND[qend].y;
  const int t0 = kv_lo_key / KVB;
  const int t1 = (max(kv_hi_key, 0) + KVB - 1) / KVB;
  const int ntiles = t1 - t0;

  constexpr int CHUNKS = KVB * D / 8;
  constexpr int PER_THR = CHUNKS / 512;
  s16x8 kreg[PER_THR], vreg[PER_THR];

  auto stage_load = [&](int tile) {
    const int kv0 = (t0 + tile) * KVB;
#pragma unroll
    for (int i = 0; i < PER_THR; ++i) {
      const int c = tid + i * 512;
      const int row = c / (D / 8);
      const int d0 = (c % (D / 8)) * 8;
      const int key = kv0 + row;
      if (key < total) {
        const long src = ((long)key * hk + kh) * D + d0;
        kreg[i] = *reinterpret_cast<const s16x8*>(K + src);
        vreg[i] = *reinterpret_cast<const s16x8*>(V + src);
      } else {
        kreg[i] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
        vreg[i] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  };

  auto stage_write = [&](int buf) {
    short* kdst = k_lds + buf * KVB * D;
    short* vdst = vt_lds + buf * KVB * D;
#pragma unroll
    for (int i = 0; i < PER_THR; ++i) {
      const int c = tid + i * 512;
      const int row = c / (D / 8);
      const int d0 = (c % (D / 8)) * 8;
      {
        unsigned byte = row * KROW_BYTES + d0 * 2;
        byte ^= (unsigned)((row & 7) << 4);
        *reinterpret_cast<s16x8*>(
            reinterpret_cast<char*>(kdst) + byte) = kreg[i];
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned byte = (unsigned)(d0 + j) * (KVB * 2) + row * 2;
        byte ^= (unsigned)((((d0 + j) >> 3) & 7) << 4);
        *reinterpret_cast<short*>(
            reinterpret_cast<char*>(vdst) + byte) = vreg[i][j];
      }
    }
  };

  float m_run = -INFINITY;
  float l_run = 0.f;
  f32x16_ oacc[NA];
#pragma unroll
  for (int a = 0; a < NA; ++a) oacc[a] = f32x16_(0.f);

  if (ntiles > 0) {
    stage_load(0);
    stage_write(0);
    if (ntiles > 1) stage_load(1);
  }
  __syncthreads();

  for (int tile = 0; tile < ntiles; ++tile) {
    const int buf = tile & 1;
    const int kv0 = (t0 + tile) * KVB;
    // wave skip: tile before the wave's first sequence, after its last
    // sequence, or beyond the causal diagonal
    bool wave_active = (kv0 + KVB > wb0.x) && (kv0 < wb31.y);
    if (CAUSAL && kv0 > q0 + 31) wave_active = false;

    if (wave_active) {
      const short* kbuf = k_lds + buf * KVB * D;
      const short* vbuf = vt_lds + buf * KVB * D;
      f32x16_ p[2];
      p[0] = f32x16_(0.f);
      p[1] = f32x16_(0.f);
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
        for (int t = 0; t < NT; ++t) {
          const int row = kb * 32 + col;
          unsigned byte = (unsigned)row * KROW_BYTES + (t * 16 + hi * 8) * 2;
          byte ^= (unsigned)((row & 7) << 4);
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(kbuf) + byte);
          p[kb] = AttnElem<FP16>::mfma(kf, qfrag[t], p[kb]);
        }
      }
      // full = every (row, key) of the wave inside one sequence span:
      // keys >= the LARGEST row start (wb31.x), < the SMALLEST row end
      // (wb0.y), and <= the smallest q row for causal
      const bool tile_full =
          (kv0 >= wb31.x) && (kv0 + KVB <= wb0.y) &&
          (!CAUSAL || kv0 + KVB - 1 <= q0);
      float pmax = -INFINITY;
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float s = p[kb][r] * scale;
          if (!tile_full) {
            const int key = kv0 + kb * 32 + CROW(r, hi);
            bool valid = (key >= myb.x) && (key < myb.y);
            if (CAUSAL) valid &= (key <= qrow);
            s = valid ? s : -INFINITY;
          }
          p[kb][r] = s;
          pmax = fmaxf(pmax, s);
        }
      }
      pmax = fmaxf(pmax, __shfl_xor(pmax, 32, 64));
      const float m_new = fmaxf(m_run, pmax);
      const float m_use = (m_new == -INFINITY) ? 0.f : m_new;
      const float alpha =
          (m_run == -INFINITY) ? 0.f : __expf(m_run - m_use);
      m_run = m_new;
      float lsum = 0.f;
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float e = __expf(p[kb][r] - m_use);
          p[kb][r] = e;
          lsum += e;
        }
      }
      lsum += __shfl_xor(lsum, 32, 64);
      l_run = l_run * alpha + lsum;
#pragma unroll
      for (int a = 0; a < NA; ++a) {
#pragma unroll
        for (int r = 0; r < 16; ++r) oacc[a][r] *= alpha;
      }
      unsigned pb[4][4];
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
        for (int tp = 0; tp < 2; ++tp) {
#pragma unroll
          for (int u = 0; u < 2; ++u) {
            const int r = 2 * u + 8 * tp;
            unsigned va = AttnElem<FP16>::cvt_pk(p[kb][r], p[kb][r + 1]);
            unsigned vb = AttnElem<FP16>::cvt_pk(p[kb][r + 4],
                                                 p[kb][r + 5]);
            auto sw = __builtin_amdgcn_permlane32_swap(va, vb, false, false);
            pb[kb * 2 + tp][u] = sw[0];
            pb[kb * 2 + tp][u + 2] = sw[1];
          }
        }
      }
#pragma unroll
      for (int a = 0; a < NA; ++a) {
#pragma unroll
        for (int st = 0; st < 4; ++st) {
          const int d = a * 32 + col;
          unsigned byte = (unsigned)d * (KVB * 2) + (st * 16 + hi * 8) * 2;
          byte ^= (unsigned)(((d >> 3) & 7) << 4);
          bf16x8 vf = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(vbuf) + byte);
          bf16x8 pf = *reinterpret_cast<const bf16x8*>(&pb[st][0]);
          oacc[a] = AttnElem<FP16>::mfma(vf, pf, oacc[a]);
        }
      }
    }

    if (tile + 1 < ntiles) {
      stage_write(buf ^ 1);
      if (tile + 2 < ntiles) stage_load(tile + 2);
    }
    __syncthreads();
  }

  if (qrow < total) {
    const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
    const long obase = ((long)qrow * hq + h) * D;
#pragma unroll
    for (int a = 0; a < NA; ++a) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = a * 32 + CROW(r, hi);
        O[obase + d] = AttnElem<FP16>::from_f32(oacc[a][r] * inv_l);
      }
    }
    if (hi == 0) {
      LSE[(long)h * total + qrow] =
          (l_run > 0.f) ? m_run + __logf(l_run) : -INFINITY;
    }
  }
}

// ---------------------------------------------------------------------------
// 2. backward preprocess: delta = rowsum(dO * O)   [hq, total] fp32
// ---------------------------------------------------------------------------
template <bool FP16>
__global__ void fa_vl_preprocess_kernel(const short* __restrict__ dO,
                                        const short* __restrict__ O,
                                        float* __restrict__ delta, int total,
                                        int hq, int D) {
  const long row = (long)blockIdx.x * (blockDim.x / WAVE) +
                   (threadIdx.x / WAVE);
  const int lane = threadIdx.x % WAVE;
  const long rows = (long)total * hq;
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
    const long s = row / hq;
    delta[(long)h * total + s] = acc;
  }
}

// ---------------------------------------------------------------------------
// 3. dK/dV kernel
// ---------------------------------------------------------------------------
template <int D, bool CAUSAL, bool FP16>
__global__ __launch_bounds__(256, 2)
void fa_vl_dkv_kernel(const short* __restrict__ dOut,
                      const short* __restrict__ Q,
                      const short* __restrict__ K,
                      const short* __restrict__ V,
                      const float* __restrict__ LSE,
                      const float* __restrict__ DELTA,
                      short* __restrict__ dK, short* __restrict__ dV,
                      const int2* __restrict__ BND, int total, int hq,
                      int hk, float scale) {
  constexpr int NT = D / 16;
  constexpr int NA = D / 32;
  constexpr int QT = 32;
  constexpr int KVWG = 128;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* q_lds = reinterpret_cast<short*>(smem);
  short* do_lds = q_lds + QT * D;
  short* v_lds = do_lds + QT * D;
  float* lse_lds = reinterpret_cast<float*>(v_lds + KVWG * D);
  float* del_lds = lse_lds + QT;
  int2* bnd_lds = reinterpret_cast<int2*>(del_lds + QT);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;
  const int hi = lane >> 5;

  const int kh = blockIdx.y;
  const int gqa = hq / hk;
  const int kv0wg = (int)((blockIdx.x + blockIdx.y) % gridDim.x) * KVWG;
  const int key_b = kv0wg + wid * 32;
  const int mykey = key_b + col;

  // wave scalars: the wave's keys span sequences [seq(key_b), seq(key_b+31)]
  const int2 kb0 = BND[min(key_b, total - 1)];
  const int2 kb31 = BND[min(key_b + 31, total - 1)];

  bf16x8 kfrag[NT];
  {
    const bool kv_ok = mykey < total;
    const long base = ((long)(kv_ok ? mykey : total - 1) * hk + kh) * D;
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
    for (int c = tid; c < CHUNKS; c += 256) {
      const int row = c / (D / 8);
      const int d0 = (c % (D / 8)) * 8;
      const int key = kv0wg + row;
      s16x8 vv8 = {0, 0, 0, 0, 0, 0, 0, 0};
      if (key < total)
        vv8 = *reinterpret_cast<const s16x8*>(V + ((long)key * hk + kh) * D +
                                              d0);
      unsigned byte = row * (D * 2) + d0 * 2;
      byte ^= (unsigned)((row & 7) << 4);
      *reinterpret_cast<s16x8*>(reinterpret_cast<char*>(v_lds) + byte) = vv8;
    }
    __syncthreads();
  }

  // per-lane constant address for the ds_read_b64_tr_b16 B-frag reads
  // (same derivation as flash_attn_bwd.hip DKV_TR_STEP)
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

  // q range for the whole workgroup: keys kv0wg..+127 need q rows of the
  // same sequences (and q >= key for causal)
  const int q_lo = CAUSAL ? kv0wg : BND[min(kv0wg, total - 1)].x;
  const int q_hi = BND[min(kv0wg + KVWG - 1, total - 1)].y;
  const int qt0 = q_lo / QT;
  const int qt1 = (max(q_hi, 0) + QT - 1) / QT;

  for (int gh = 0; gh < gqa; ++gh) {
    const int h = kh * gqa + gh;
    for (int qt = qt0; qt < qt1; ++qt) {
      const int q0 = qt * QT;
      {
        constexpr int CHUNKS = QT * D / 8;
        for (int c = tid; c < CHUNKS; c += 256) {
          const int row = c / (D / 8);
          const int d0 = (c % (D / 8)) * 8;
          const int qrow = q0 + row;
          s16x8 qv = {0, 0, 0, 0, 0, 0, 0, 0};
          s16x8 dv8 = {0, 0, 0, 0, 0, 0, 0, 0};
          if (qrow < total) {
            const long src = ((long)qrow * hq + h) * D + d0;
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
          const long idx = (long)h * total + min(qrow, total - 1);
          lse_lds[r] = (qrow < total) ? LSE[idx] : INFINITY;
          del_lds[r] = (qrow < total) ? DELTA[idx] : 0.f;
          bnd_lds[r] = (qrow < total) ? BND[qrow] : int2{0, 0};
        }
      }
      __syncthreads();

      // wave skip: q tile entirely outside this wave's keys' sequences /
      // causal triangle
      bool wave_active = (q0 + QT > kb0.x) && (q0 < kb31.y);
      if (CAUSAL && q0 + QT - 1 < key_b) wave_active = false;

      if (wave_active) {
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
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qrow = q0 + CROW(r, hi);
          const float lse_q = lse_lds[CROW(r, hi)];
          const float del_q = del_lds[CROW(r, hi)];
          const int2 bq = bnd_lds[CROW(r, hi)];
          bool valid = isfinite(lse_q) && (mykey >= bq.x) && (mykey < bq.y);
          if (CAUSAL) valid &= (mykey <= qrow);
          const float p = valid ? __expf(s[r] * scale - lse_q) : 0.f;
          s[r] = p;
          dp[r] = p * (dp[r] - del_q) * scale;
        }
#define VL_TR_STEP(tp_, a_)                                                \
        {                                                                  \
          attn_u32x2 ql_, qh_, dl_, dh_;                                   \
          asm volatile(                                                    \
              "ds_read_b64_tr_b16 %0, %4 offset:%c5\n\t"                 \
              "ds_read_b64_tr_b16 %1, %4 offset:%c6\n\t"                 \
              "ds_read_b64_tr_b16 %2, %4 offset:%c7\n\t"                 \
              "ds_read_b64_tr_b16 %3, %4 offset:%c8\n\t"                 \
              "s_waitcnt lgkmcnt(0)"                                       \
              : "=v"(ql_), "=v"(qh_), "=v"(dl_), "=v"(dh_)                 \
              : "v"(tr_addr),                                              \
                "i"((tp_) * 16 * D * 2 + (a_) * 64),                       \
                "i"((tp_) * 16 * D * 2 + 4 * D * 2 + ((a_) ^ 1) * 64),     \
                "i"(QT * D * 2 + (tp_) * 16 * D * 2 + (a_) * 64),          \
                "i"(QT * D * 2 + (tp_) * 16 * D * 2 + 4 * D * 2 +          \
                    ((a_) ^ 1) * 64));                                     \
          attn_u32x4 uq_ = {ql_.x, ql_.y, qh_.x, qh_.y};                   \
          attn_u32x4 ud_ = {dl_.x, dl_.y, dh_.x, dh_.y};                   \
          bf16x8 qbf = __builtin_bit_cast(bf16x8, uq_);                    \
          bf16x8 dob = __builtin_bit_cast(bf16x8, ud_);                    \
          dvacc[a_] = AttnElem<FP16>::mfma(pb, dob, dvacc[a_]);            \
          dkacc[a_] = AttnElem<FP16>::mfma(dsb, qbf, dkacc[a_]);           \
        }

#pragma unroll
        for (int tp = 0; tp < 2; ++tp) {
          unsigned pfr[4], dsfr[4];
          t12_pack_frag<AttnElem<FP16>>(s, tp, pfr);
          t12_pack_frag<AttnElem<FP16>>(dp, tp, dsfr);
          bf16x8 pb = *reinterpret_cast<const bf16x8*>(pfr);
          bf16x8 dsb = *reinterpret_cast<const bf16x8*>(dsfr);
          if (tp == 0) {
            VL_TR_STEP(0, 0);
            VL_TR_STEP(0, 1);
            if constexpr (NA > 2) {
              VL_TR_STEP(0, 2);
              VL_TR_STEP(0, 3);
            }
          } else {
            VL_TR_STEP(1, 0);
            VL_TR_STEP(1, 1);
            if constexpr (NA > 2) {
              VL_TR_STEP(1, 2);
              VL_TR_STEP(1, 3);
            }
          }
        }
#undef VL_TR_STEP
      }
      __syncthreads();
    }
  }

#pragma unroll
  for (int a = 0; a < NA; ++a) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int key = key_b + CROW(r, hi);
      const int d = a * 32 + col;
      if (key < total) {
        const long base = ((long)key * hk + kh) * D;
        dK[base + d] = AttnElem<FP16>::from_f32(dkacc[a][r]);
        dV[base + d] = AttnElem<FP16>::from_f32(dvacc[a][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 4. dQ kernel (forward structure, 128-key staged tiles)
// ---------------------------------------------------------------------------
template <int D, bool CAUSAL, bool FP16>
__global__ __launch_bounds__(512, 2)
void fa_vl_dq_kernel(const short* __restrict__ dOut,
                     const short* __restrict__ Q,
                     const short* __restrict__ K,
                     const short* __restrict__ V,
                     const float* __restrict__ LSE,
                     const float* __restrict__ DELTA,
                     short* __restrict__ dQ,
                     const int2* __restrict__ BND, int total, int hq,
                     int hk, float scale) {
  constexpr int NT = D / 16;
  constexpr int NA = D / 32;
  constexpr int KVB = 128;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* k_lds = reinterpret_cast<short*>(smem);
  short* v_lds = k_lds + KVB * D;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 31;
  const int hi = lane >> 5;

  const int h = blockIdx.y;
  const int kh = h / (hq / hk);
  const int q0wg = (int)((blockIdx.x + blockIdx.y) % gridDim.x) * 256;
  const int q0 = q0wg + wid * 32;
  const int qrow = q0 + col;

  int2 myb = BND[min(qrow, total - 1)];
  if (qrow >= total) myb = int2{0, 0};
  const int2 wb0 = BND[min(q0, total - 1)];
  const int2 wb31 = BND[min(q0 + 31, total - 1)];

  bf16x8 qfrag[NT], dofrag[NT];
  {
    const long qbase = ((long)min(qrow, total - 1) * hq + h) * D;
    const bool qvalid = (qrow < total);
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
  const bool row_ok = (qrow < total);
  const float lse_q =
      row_ok ? LSE[(long)h * total + qrow] : INFINITY;
  const float del_q =
      row_ok ? DELTA[(long)h * total + qrow] : 0.f;

  // per-lane constant for ds_read_b64_tr_b16 K B-frag reads (same
  // derivation as flash_attn_bwd.hip)
  const int m4 = (lane & 15) >> 2;
  const int dl0 = ((lane >> 4) & 1) * 16 + (lane & 3) * 4;
  const unsigned ktr_base = (unsigned)(size_t)k_lds +
      (unsigned)((hi * 8 + m4) * (D * 2)) +
      (((unsigned)(dl0 * 2)) ^ ((unsigned)(m4 << 4)));

  f32x16 dqacc[NA];
#pragma unroll
  for (int a = 0; a < NA; ++a) dqacc[a] = f32x16(0.f);

  const int qend = min(q0wg + 255, total - 1);
  const int kv_lo_key = BND[min(q0wg, total - 1)].x;
  const int kv_hi_key = CAUSAL ? (qend + 1) : BND[qend].y;
  const int t0 = kv_lo_key / KVB;
  const int t1 = (max(kv_hi_key, 0) + KVB - 1) / KVB;

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
        if (key < total) {
          const long src = ((long)key * hk + kh) * D + d0;
          kv8 = *reinterpret_cast<const s16x8*>(K + src);
          vv8 = *reinterpret_cast<const s16x8*>(V + src);
        }
        unsigned byte = row * (D * 2) + d0 * 2;
        byte ^= (unsigned)((row & 7) << 4);
        *reinterpret_cast<s16x8*>(reinterpret_cast<char*>(k_lds) + byte) =
            kv8;
        *reinterpret_cast<s16x8*>(reinterpret_cast<char*>(v_lds) + byte) =
            vv8;
      }
    }
    __syncthreads();

#pragma unroll 1
    for (int half = 0; half < KVB / 64; ++half) {
    const int kv0h = kv0 + half * 64;
    bool wave_active = kv0h < kv_hi_key &&
        (kv0h + 64 > wb0.x) && (kv0h < wb31.y);
    if (CAUSAL && kv0h > q0 + 31) wave_active = false;

    if (wave_active) {
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
          const int row = half * 64 + kb * 32 + col;
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
      const bool lane_ok = row_ok && isfinite(lse_q);
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int key = kv0h + kb * 32 + CROW(r, hi);
          bool valid = lane_ok && (key >= myb.x) && (key < myb.y);
          if (CAUSAL) valid &= (key <= qrow);
          const float p = valid ? __expf(s[kb][r] * scale - lse_q) : 0.f;
          dp[kb][r] = p * (dp[kb][r] - del_q) * scale;
        }
      }
      const unsigned ktr_addr = ktr_base + (unsigned)(half * 64 * D * 2);

#define VLQ_TR_STEP(kb_, tp_, a_)                                          \
      {                                                                    \
        attn_u32x2 kl_, kh_;                                               \
        asm volatile(                                                      \
            "ds_read_b64_tr_b16 %0, %2 offset:%c3\n\t"                   \
            "ds_read_b64_tr_b16 %1, %2 offset:%c4\n\t"                   \
            "s_waitcnt lgkmcnt(0)"                                         \
            : "=v"(kl_), "=v"(kh_)                                         \
            : "v"(ktr_addr),                                               \
              "i"((kb_) * 32 * D * 2 + (tp_) * 16 * D * 2 + (a_) * 64),    \
              "i"((kb_) * 32 * D * 2 + (tp_) * 16 * D * 2 + 4 * D * 2 +    \
                  ((a_) ^ 1) * 64));                                       \
        attn_u32x4 uk_ = {kl_.x, kl_.y, kh_.x, kh_.y};                     \
        bf16x8 ktb = __builtin_bit_cast(bf16x8, uk_);                      \
        dqacc[a_] = AttnElem<FP16>::mfma(dsb, ktb, dqacc[a_]);             \
      }
#define VLQ_TR_TP(kb_, tp_)                                                \
      {                                                                    \
        unsigned dsfr[4];                                                  \
        t12_pack_frag<AttnElem<FP16>>(dp[kb_], tp_, dsfr);                 \
        bf16x8 dsb = *reinterpret_cast<const bf16x8*>(dsfr);               \
        VLQ_TR_STEP(kb_, tp_, 0);                                          \
        VLQ_TR_STEP(kb_, tp_, 1);                                          \
        if constexpr (NA > 2) {                                            \
          VLQ_TR_STEP(kb_, tp_, 2);                                        \
          VLQ_TR_STEP(kb_, tp_, 3);                                        \
        }                                                                  \
      }
      VLQ_TR_TP(0, 0);
      VLQ_TR_TP(0, 1);
      VLQ_TR_TP(1, 0);
      VLQ_TR_TP(1, 1);
#undef VLQ_TR_TP
#undef VLQ_TR_STEP
    }
    }  // half
    __syncthreads();
  }

#pragma unroll
  for (int a = 0; a < NA; ++a) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int q_r = q0 + CROW(r, hi);
      const int d = a * 32 + col;
      if (q_r < total) {
        const long obase = ((long)q_r * hq + h) * D;
        dQ[obase + d] = AttnElem<FP16>::from_f32(dqacc[a][r]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

std::vector<torch::Tensor> fa_varlen_forward(torch::Tensor q, torch::Tensor k,
                                             torch::Tensor v,
                                             torch::Tensor bounds,
                                             double softmax_scale,
                                             bool causal) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
              v.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16 ||
                  q.scalar_type() == torch::kHalf,
              "fa_varlen_forward: bf16/fp16 only (CDNA4 MFMA path)");
  const bool fp16 = q.scalar_type() == torch::kHalf;
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3, "packed [total, h, d] expected");
  const int total = q.size(0), hq = q.size(1), D = q.size(2);
  const int hk = k.size(1);
  TORCH_CHECK(k.size(0) == total,
              "fa_varlen_forward requires identically packed q/k");
  TORCH_CHECK(D == 64 || D == 128);
  TORCH_CHECK(hq % hk == 0);
  TORCH_CHECK(bounds.scalar_type() == torch::kInt32 && bounds.is_cuda() &&
              bounds.is_contiguous() && bounds.numel() == 2 * (long)total,
              "bounds must be int32 [total, 2] on device");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({hq, total}, q.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid((total + 255) / 256, hq), block(512);

#define LAUNCH_F(DD, C)                                                      \
  if (fp16)                                                                  \
    hipLaunchKernelGGL((fa_vl_fwd_kernel<DD, C, true>), grid, block,         \
                       4 * 64 * DD * 2, stream, (const short*)q.data_ptr(),  \
                       (const short*)k.data_ptr(),                           \
                       (const short*)v.data_ptr(), (short*)o.data_ptr(),     \
                       lse.data_ptr<float>(),                                \
                       (const int2*)bounds.data_ptr<int>(), total, hq, hk,   \
                       (float)softmax_scale);                                \
  else                                                                       \
    hipLaunchKernelGGL((fa_vl_fwd_kernel<DD, C, false>), grid, block,        \
                     4 * 64 * DD * 2, stream, (const short*)q.data_ptr(),    \
                     (const short*)k.data_ptr(),                             \
                     (const short*)v.data_ptr(), (short*)o.data_ptr(),       \
                     lse.data_ptr<float>(),                                  \
                     (const int2*)bounds.data_ptr<int>(), total, hq, hk,     \
                     (float)softmax_scale)
  if (D == 128) { if (causal) LAUNCH_F(128, true); else LAUNCH_F(128, false); }
  else { if (causal) LAUNCH_F(64, true); else LAUNCH_F(64, false); }
#undef LAUNCH_F
  HIP_CHECK_LAST();
  return {o, lse};
}

std::vector<torch::Tensor> fa_varlen_backward(
    torch::Tensor dout, torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor out, torch::Tensor lse, torch::Tensor bounds,
    double softmax_scale, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
              v.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16 ||
              q.scalar_type() == torch::kHalf);
  const bool fp16 = q.scalar_type() == torch::kHalf;
  dout = dout.contiguous();
  const int total = q.size(0), hq = q.size(1), D = q.size(2);
  const int hk = k.size(1);
  TORCH_CHECK(D == 64 || D == 128);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto delta = torch::empty({hq, total},
                            q.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();

  {
    const long rows = (long)total * hq;
    const int rpb = 256 / WAVE;
    dim3 g((rows + rpb - 1) / rpb), bl(256);
    if (fp16)
      hipLaunchKernelGGL(fa_vl_preprocess_kernel<true>, g, bl, 0, stream,
                         (const short*)dout.data_ptr(),
                         (const short*)out.data_ptr(),
                         delta.data_ptr<float>(), total, hq, D);
    else
      hipLaunchKernelGGL(fa_vl_preprocess_kernel<false>, g, bl, 0, stream,
                         (const short*)dout.data_ptr(),
                         (const short*)out.data_ptr(),
                         delta.data_ptr<float>(), total, hq, D);
  }

  dim3 gkv((total + 127) / 128, hk), bkv(256);
  dim3 gq((total + 255) / 256, hq), bq(512);

#define LAUNCH_B(DD, C)                                                      \
  do {                                                                       \
    const int lds_kv =                                                       \
        (2 * 32 * DD + 128 * DD) * 2 + 2 * 32 * 4 + 32 * 8;                  \
    const int lds_q = 2 * 128 * DD * 2;                                      \
    if (fp16)                                                              \
      hipLaunchKernelGGL((fa_vl_dkv_kernel<DD, C, true>), gkv, bkv, lds_kv, \
                       stream,                                              \
                       (const short*)dout.data_ptr(),                       \
                       (const short*)q.data_ptr(),                          \
                       (const short*)k.data_ptr(),                          \
                       (const short*)v.data_ptr(), lse.data_ptr<float>(),   \
                       delta.data_ptr<float>(), (short*)dk.data_ptr(),      \
                       (short*)dv.data_ptr(),                               \
                       (const int2*)bounds.data_ptr<int>(), total, hq, hk,  \
                       (float)softmax_scale);                               \
    else                                                                    \
      hipLaunchKernelGGL((fa_vl_dkv_kernel<DD, C, false>), gkv, bkv,        \
                       lds_kv, stream,                                      \
                       (const short*)dout.data_ptr(),                        \
                       (const short*)q.data_ptr(),                           \
                       (const short*)k.data_ptr(),                           \
                       (const short*)v.data_ptr(), lse.data_ptr<float>(),    \
                       delta.data_ptr<float>(), (short*)dk.data_ptr(),       \
                       (short*)dv.data_ptr(),                                \
                       (const int2*)bounds.data_ptr<int>(), total, hq, hk,   \
                       (float)softmax_scale);                                \
    if (fp16)                                                              \
      hipLaunchKernelGGL((fa_vl_dq_kernel<DD, C, true>), gq, bq, lds_q,     \
                       stream,                                              \
                       (const short*)dout.data_ptr(),                       \
                       (const short*)q.data_ptr(),                          \
                       (const short*)k.data_ptr(),                          \
                       (const short*)v.data_ptr(), lse.data_ptr<float>(),   \
                       delta.data_ptr<float>(), (short*)dq.data_ptr(),      \
                       (const int2*)bounds.data_ptr<int>(), total, hq, hk,  \
                       (float)softmax_scale);                               \
    else                                                                    \
      hipLaunchKernelGGL((fa_vl_dq_kernel<DD, C, false>), gq, bq, lds_q,    \
                       stream,                                              \
                       (const short*)dout.data_ptr(),                        \
                       (const short*)q.data_ptr(),                           \
                       (const short*)k.data_ptr(),                           \
                       (const short*)v.data_ptr(), lse.data_ptr<float>(),    \
                       delta.data_ptr<float>(), (short*)dq.data_ptr(),       \
                       (const int2*)bounds.data_ptr<int>(), total, hq, hk,   \
                       (float)softmax_scale);                                \
  } while (0)
  if (D == 128) { if (causal) LAUNCH_B(128, true); else LAUNCH_B(128, false); }
  else { if (causal) LAUNCH_B(64, true); else LAUNCH_B(64, false); }
#undef LAUNCH_B
  HIP_CHECK_LAST();
  return {dq, dk, dv};
}
