// Full-featured flash-attention FORWARD (alibi + dropout), separate
// translation unit: the vanilla hot path (flash_attn_fwd.hip) stays at its
// tuned register budget; these variants tolerate spills (guide rule 19:
// co-compiled template variants perturb each other's codegen).
// Flash-attention v2 FORWARD for gfx950 (CDNA4 MFMA).
//
// Structure (per the CDNA4 attention ladder in the MI355X guides):
//   - workgroup = 8 waves (512 threads) = 256 Q rows; each wave owns a
//     32-row Q block. Q stays in registers (D/16 bf16x8 fragments/lane).
//   - KV tile = 64 keys staged in LDS, double-buffered: K row-major with the
//     (row&7)<<4 XOR byte swizzle (conflict-free ds_read_b128 column-slice
//     reads), V stored TRANSPOSED [D][64] with the (d&7)<<4 swizzle.
//   - QK^T is computed SWAPPED: mfma_f32_32x32x16_bf16(A=K, B=Q) so each
//     lane holds the score row of ONE q (col = lane&31): the softmax row
//     reduction is 31 in-lane ops + one cross-half __shfl_xor(.,32).
//   - online softmax in registers (m, l per lane); P is converted to bf16
//     with v_cvt_pk_bf16_f32 and redistributed across lane halves with
//     permlane32_swap, landing EXACTLY in the PV mfma's B-fragment layout.
//   - PV: mfma(A=V^T fragment from LDS, B=P) accumulating OUT^T[d][q];
//     epilogue normalizes by 1/l and stores out + logsumexp.
//
// Covers: causal (bottom-right aligned), GQA (h_kv | h), sliding window,
// per-batch varlen (q_lens/k_lens), D in {64, 128}. bf16 only.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "attn_common.h"

typedef float f32x16_ __attribute__((ext_vector_type(16)));
#define cvt_pk_bf16 attn_cvt_pk_bf16  // bf16 path alias

// HAS_EXTRA: alibi and/or dropout active (kept out of the
// register budget of the vanilla path)
template <int D, bool CAUSAL, bool HAS_WINDOW, bool HAS_LENS, bool HAS_EXTRA,
          bool FP16>
__global__ __launch_bounds__(512, 2)
void fa_fwd_kernel_x(const short* __restrict__ Q, const short* __restrict__ K,
                   const short* __restrict__ V, short* __restrict__ O,
                   float* __restrict__ LSE, int b_, int sq, int sk, int hq,
                   int hk, float scale, int wl, int wr,
                   const int* __restrict__ q_lens,
                   const int* __restrict__ k_lens,
                   const float* __restrict__ alibi, float p_drop,
                   unsigned long long rng_seed) {
  constexpr int NT = D / 16;   // QK^T k-steps (d slices of 16)
  constexpr int NA = D / 32;   // PV output accs (d blocks of 32)
  constexpr int KVB = 64;      // keys per tile
  constexpr int KROW_BYTES = D * 2;

  // LDS: K [64][D] + V^T [D][64], double buffered
  extern __shared__ __attribute__((aligned(16))) char smem[];
  short* k_lds = reinterpret_cast<short*>(smem);                // 2*64*D
  short* vt_lds = reinterpret_cast<short*>(smem) + 2 * KVB * D; // 2*D*64

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;          // 0..7
  const int col = lane & 31;         // q row within wave block / d col
  const int hi = lane >> 5;          // lane half

  const int h = blockIdx.y % hq;
  const int b = blockIdx.y / hq;
  const int kh = h / (hq / hk);
  // tile<->XCD decorrelation: the dispatcher places block i on XCD i%8 and
  // blockIdx.x varies fastest, so with causal tile-skipping a direct
  // qtile=blockIdx.x maps every heavy (diagonal) workgroup onto the same
  // XCD; shifting by blockIdx.y balances the causal triangle (bijective
  // per y-slice)
  const int q0wg = (int)((blockIdx.x + blockIdx.y) % gridDim.x) * 256;
  const int q0 = q0wg + wid * 32;        // this wave's q block
  const int qrow = q0 + col;             // this lane's q row
  const int shift = sk - sq;             // bottom-right causal alignment

  int klimit = sk;
  int qlimit = sq;
  if (HAS_LENS) {
    if (k_lens != nullptr) klimit = min(klimit, k_lens[b]);
    if (q_lens != nullptr) qlimit = min(qlimit, q_lens[b]);
  }
  const float slope =
      (HAS_EXTRA && alibi != nullptr) ? alibi[h] : 0.f;
  const bool has_alibi = HAS_EXTRA && alibi != nullptr;
  const bool has_drop = HAS_EXTRA && p_drop > 0.f;
  const float inv_keep = HAS_EXTRA ? 1.f / (1.f - p_drop) : 1.f;
  const unsigned thr24 =
      HAS_EXTRA ? (unsigned)(p_drop * 16777216.f) : 0u;

  // ---- load Q fragments (pre-scaled) -------------------------------------
  // B-frag for swapped QK^T: lane holds Q[qrow][t*16 + hi*8 + j], j=0..7
  bf16x8 qfrag[NT];
  {
    const long qbase = ((long)(b * sq + min(qrow, sq - 1)) * hq + h) * D;
    const bool qvalid = (qrow < sq);
#pragma unroll
    for (int t = 0; t < NT; ++t) {
      s16x8 v = qvalid
          ? *reinterpret_cast<const s16x8*>(Q + qbase + t * 16 + hi * 8)
          : s16x8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
      for (int j = 0; j < 8; ++j) qfrag[t][j] = v[j];
    }
  }

  // ---- KV tile range -----------------------------------------------------
  int kv_hi_key = klimit;  // exclusive
  if (CAUSAL) kv_hi_key = min(kv_hi_key, q0wg + 255 + shift + 1);
  int kv_lo_key = 0;
  if (HAS_WINDOW && wl >= 0) kv_lo_key = max(0, q0wg + shift - wl);
  const int t0 = kv_lo_key / KVB;
  const int t1 = (kv_hi_key + KVB - 1) / KVB;  // exclusive
  const int ntiles = t1 - t0;

  // ---- staging helpers (T14 async split: issue loads early, write LDS
  // late so HBM latency hides under the previous tile's compute) ----------
  const long kbase = ((long)b * sk * hk + kh) * D;  // + key*hk*D + d
  constexpr int CHUNKS = KVB * D / 8;      // 16B chunks in tile
  constexpr int PER_THR = CHUNKS / 512;    // chunks per thread
  s16x8 kreg[PER_THR], vreg[PER_THR];

  auto stage_load = [&](int tile) {
    const int kv0 = (t0 + tile) * KVB;
#pragma unroll
    for (int i = 0; i < PER_THR; ++i) {
      const int c = tid + i * 512;
      const int row = c / (D / 8);
      const int d0 = (c % (D / 8)) * 8;
      const int key = kv0 + row;
      if (key < sk) {
        const long src = kbase + (long)key * hk * D + d0;
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
      // K: row-major with (row&7)<<4 byte-XOR swizzle
      {
        unsigned byte = row * KROW_BYTES + d0 * 2;
        byte ^= (unsigned)((row & 7) << 4);
        *reinterpret_cast<s16x8*>(
            reinterpret_cast<char*>(kdst) + byte) = kreg[i];
      }
      // V^T: [D][64] with (d&7)<<4 swizzle, scalar scatter
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned byte = (unsigned)(d0 + j) * (KVB * 2) + row * 2;
        byte ^= (unsigned)((((d0 + j) >> 3) & 7) << 4);
        *reinterpret_cast<short*>(
            reinterpret_cast<char*>(vdst) + byte) = vreg[i][j];
      }
    }
  };

  // ---- online softmax state ---------------------------------------------
  float m_run = -INFINITY;
  float l_run = 0.f;
  f32x16_ oacc[NA];
#pragma unroll
  for (int a = 0; a < NA; ++a) oacc[a] = f32x16_(0.f);

  if (ntiles > 0) {
    stage_load(0);
    stage_write(0);
    if (ntiles > 1) stage_load(1);  // in flight under tile 0's compute
  }
  __syncthreads();

  for (int tile = 0; tile < ntiles; ++tile) {
    const int buf = tile & 1;
    const int kv0 = (t0 + tile) * KVB;
    // wave-level skip: tile entirely outside this wave's causal/window span
    bool wave_active = true;
    if (CAUSAL && kv0 > q0 + 31 + shift) wave_active = false;
    if (HAS_WINDOW && wl >= 0 && kv0 + KVB <= q0 + shift - wl)
      wave_active = false;
    if (qrow >= qlimit && __all(qrow >= qlimit)) {
      // whole wave beyond valid q rows: nothing to compute
    }

    if (wave_active) {
      const short* kbuf = k_lds + buf * KVB * D;
      const short* vbuf = vt_lds + buf * KVB * D;
      // ---- QK^T: 2 key sub-blocks x NT d-steps --------------------------
      f32x16_ p[2];
      p[0] = f32x16_(0.f);
      p[1] = f32x16_(0.f);
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
        for (int t = 0; t < NT; ++t) {
          // A-frag: K[key = kb*32 + col][d = t*16 + hi*8 + j]
          const int row = kb * 32 + col;
          unsigned byte = (unsigned)row * KROW_BYTES + (t * 16 + hi * 8) * 2;
          byte ^= (unsigned)((row & 7) << 4);
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(kbuf) + byte);
          p[kb] = AttnElem<FP16>::mfma(kf, qfrag[t], p[kb]);
        }
      }
      // ---- mask + scale --------------------------------------------------
      // "full" = no per-element masking needed for ANY q row of this wave:
      // causal bound uses the wave's SMALLEST q (q0), window-left bound its
      // LARGEST q (q0+31), window-right its smallest.
      const bool tile_full = !has_alibi &&
          (!CAUSAL || kv0 + KVB - 1 <= q0 + shift) &&
          (kv0 + KVB <= klimit) &&
          (!HAS_WINDOW || wl < 0 || kv0 >= q0 + 31 + shift - wl) &&
          (!HAS_WINDOW || wr < 0 || CAUSAL ||
           kv0 + KVB - 1 <= q0 + shift + wr);
      float pmax = -INFINITY;
#pragma unroll
      for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float s = p[kb][r] * scale;
          if (!tile_full) {
            const int key = kv0 + kb * 32 + CROW(r, hi);
            if (HAS_EXTRA && has_alibi)
              s -= slope * fabsf((float)(qrow + shift - key));
            bool valid = key < klimit;
            if (CAUSAL) valid &= (key <= qrow + shift);
            if (HAS_WINDOW && wl >= 0) valid &= (key >= qrow + shift - wl);
            if (HAS_WINDOW && wr >= 0 && !CAUSAL)
              valid &= (key <= qrow + shift + wr);
            s = valid ? s : -INFINITY;
          }
          p[kb][r] = s;
          pmax = fmaxf(pmax, s);
        }
      }
      pmax = fmaxf(pmax, __shfl_xor(pmax, 32, 64));
      // ---- online rescale ------------------------------------------------
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
          float e = __expf(p[kb][r] - m_use);  // -inf -> 0
          p[kb][r] = e;
          lsum += e;
        }
      }
      lsum += __shfl_xor(lsum, 32, 64);
      l_run = l_run * alpha + lsum;
      if (HAS_EXTRA && has_drop) {
        // dropout AFTER the softmax statistics: lse is unaffected
#pragma unroll
        for (int kb = 0; kb < 2; ++kb) {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int key = kv0 + kb * 32 + CROW(r, hi);
            const unsigned long long idx =
                (((unsigned long long)(b * hq + h) * sq + qrow)) *
                    (unsigned long long)sk + key;
            p[kb][r] = attn_dropout_keep(rng_seed, idx, thr24)
                           ? p[kb][r] * inv_keep : 0.f;
          }
        }
      }
#pragma unroll
      for (int a = 0; a < NA; ++a) {
#pragma unroll
        for (int r = 0; r < 16; ++r) oacc[a][r] *= alpha;
      }
      // ---- P -> bf16 B-fragments (cvt_pk + permlane32_swap) -------------
      // step (kb, tp): B-frag covering keys [kv0+kb*32+16*tp, +16)
      unsigned pb[4][4];  // [step][u32 slot]
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
      // ---- PV: OUT^T[d][q] += V^T x P -----------------------------------
#pragma unroll
      for (int a = 0; a < NA; ++a) {
#pragma unroll
        for (int st = 0; st < 4; ++st) {
          // A-frag: V^T[d = a*32 + col][key = st*16 + hi*8 + j]
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
      // write tile+1 (loaded one iteration ago) into the other buffer --
      // safe: buf^1 was last read during tile-1, behind a barrier
      stage_write(buf ^ 1);
      if (tile + 2 < ntiles) stage_load(tile + 2);
    }
    __syncthreads();
  }

  // ---- epilogue ----------------------------------------------------------
  const bool qvalid = (qrow < qlimit) && (qrow < sq);
  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
  if (qvalid) {
    const long obase = ((long)(b * sq + qrow) * hq + h) * D;
#pragma unroll
    for (int a = 0; a < NA; ++a) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = a * 32 + CROW(r, hi);
        O[obase + d] = AttnElem<FP16>::from_f32(oacc[a][r] * inv_l);
      }
    }
    if (hi == 0) {
      LSE[((long)b * hq + h) * sq + qrow] =
          (l_run > 0.f) ? m_run + __logf(l_run) : -INFINITY;
    }
  } else if (qrow < sq) {
    // rows masked away by q_lens: zero output, zero lse (reference behavior)
    const long obase = ((long)(b * sq + qrow) * hq + h) * D;
#pragma unroll
    for (int a = 0; a < NA; ++a) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = a * 32 + CROW(r, hi);
        O[obase + d] = 0;
      }
    }
    if (hi == 0) LSE[((long)b * hq + h) * sq + qrow] = 0.f;
  }
}

// ---------------------------------------------------------------------------
// host wrapper
// ---------------------------------------------------------------------------

template <int D, bool FP16>
static void launch_fa_fwd_x(const torch::Tensor& q, const torch::Tensor& k,
                          const torch::Tensor& v, torch::Tensor& o,
                          torch::Tensor& lse, float scale, bool causal,
                          int wl, int wr, const torch::Tensor& q_lens,
                          const torch::Tensor& k_lens,
                          const torch::Tensor& alibi, float p_drop,
                          unsigned long long rng_seed, hipStream_t stream) {
  const int b = q.size(0), sq = q.size(1), hq = q.size(2);
  const int sk = k.size(1), hk = k.size(2);
  const bool has_window = (wl >= 0 || wr >= 0);
  const bool has_lens = q_lens.numel() > 0 || k_lens.numel() > 0;
  dim3 grid((sq + 255) / 256, b * hq);
  dim3 block(512);
  const int lds = 4 * 64 * D * 2;  // K + V^T, double buffered
  const int* qlp = q_lens.numel() ? q_lens.data_ptr<int>() : nullptr;
  const int* klp = k_lens.numel() ? k_lens.data_ptr<int>() : nullptr;
  const float* alp = alibi.numel() ? alibi.data_ptr<float>() : nullptr;

#define LAUNCH_CASE(C, W, L, DR)                                            \
  if (causal == C && has_window == W && has_lens == L && has_drop == DR) {   \
    hipLaunchKernelGGL((fa_fwd_kernel_x<D, C, W, L, DR, FP16>), grid,       \
                       block, lds, stream, (const short*)q.data_ptr(),       \
                       (const short*)k.data_ptr(),                           \
                       (const short*)v.data_ptr(), (short*)o.data_ptr(),     \
                       lse.data_ptr<float>(), b, sq, sk, hq, hk, scale, wl,  \
                       wr, qlp, klp, alp, p_drop, rng_seed);                 \
    return;                                                                  \
  }
  const bool has_drop = true;  // extra TU: alibi/dropout active
  LAUNCH_CASE(false, false, false, true)
  LAUNCH_CASE(false, false, true, true)
  LAUNCH_CASE(false, true, false, true)
  LAUNCH_CASE(false, true, true, true)
  LAUNCH_CASE(true, false, false, true)
  LAUNCH_CASE(true, false, true, true)
  LAUNCH_CASE(true, true, false, true)
  LAUNCH_CASE(true, true, true, true)
#undef LAUNCH_CASE
}

std::vector<torch::Tensor> fa_forward_extra(torch::Tensor q, torch::Tensor k,
                                      torch::Tensor v, double softmax_scale,
                                      bool causal, long wl, long wr,
                                      torch::Tensor q_lens,
                                      torch::Tensor k_lens,
                                      torch::Tensor alibi_slopes,
                                      double p_drop, long rng_seed) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous() &&
              v.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16 ||
                  q.scalar_type() == torch::kHalf,
              "fa_forward: bf16/fp16 only (CDNA4 MFMA path)");
  const bool fp16 = q.scalar_type() == torch::kHalf;
  const int D = q.size(3);
  TORCH_CHECK(D == 64 || D == 128, "fa_forward: head_dim must be 64 or 128");
  TORCH_CHECK(q.size(2) % k.size(2) == 0);
  auto o = torch::empty_like(q);
  auto lse = torch::empty({q.size(0), q.size(2), q.size(1)},
                          q.options().dtype(torch::kFloat32));
  auto ql = q_lens.numel() ? q_lens.to(q.device(), torch::kInt32)
                           : q_lens;
  auto kl = k_lens.numel() ? k_lens.to(q.device(), torch::kInt32)
                           : k_lens;
  auto al = alibi_slopes.numel()
                ? alibi_slopes.to(q.device(), torch::kFloat32).contiguous()
                : alibi_slopes;
  TORCH_CHECK(al.numel() == 0 || al.numel() == q.size(2),
              "alibi_slopes must be [num_heads]");
  auto stream = at::hip::getCurrentHIPStream();
  if (D == 128) {
    if (fp16)
      launch_fa_fwd_x<128, true>(q, k, v, o, lse, (float)softmax_scale,
                                 causal, (int)wl, (int)wr, ql, kl, al,
                                 (float)p_drop, (unsigned long long)rng_seed,
                                 stream);
    else
      launch_fa_fwd_x<128, false>(q, k, v, o, lse, (float)softmax_scale,
                                  causal, (int)wl, (int)wr, ql, kl, al,
                                  (float)p_drop,
                                  (unsigned long long)rng_seed, stream);
  } else {
    if (fp16)
      launch_fa_fwd_x<64, true>(q, k, v, o, lse, (float)softmax_scale,
                                causal, (int)wl, (int)wr, ql, kl, al,
                                (float)p_drop, (unsigned long long)rng_seed,
                                stream);
    else
      launch_fa_fwd_x<64, false>(q, k, v, o, lse, (float)softmax_scale,
                                 causal, (int)wl, (int)wr, ql, kl, al,
                                 (float)p_drop, (unsigned long long)rng_seed,
                                 stream);
  }
  HIP_CHECK_LAST();
  return {o, lse};
}
