// Memory-bound fused elementwise/normalization kernels for gfx950:
// RMSNorm fwd/bwd, RoPE (fused q+k), SwiGLU fwd/bwd.
//
// All HBM-bound: bf16x8 (16 B/lane) vectorized loads/stores, grid-stride
// loops capped at ~2048 workgroups (guide G11/G13). fp32 math throughout.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"
#include "attn_common.h"

// ---------------------------------------------------------------------------
// RMSNorm
// ---------------------------------------------------------------------------
// one 256-thread block per row (4 waves); H assumed multiple of 8.

template <typename T>
struct VecIO;

// bf16 path: 8 elements per 16B
template <bool FP16>
__global__ void rmsnorm_fwd_kernel(const short* __restrict__ x,
                                   const short* __restrict__ w,
                                   short* __restrict__ y,
                                   float* __restrict__ inv_rms, int rows,
                                   int H, float eps) {
  __shared__ float red[16];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short* xr = x + (long)row * H;
    short* yr = y + (long)row * H;
    float ss = 0.f;
    for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
      s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = AttnElem<FP16>::to_f32(v[j]);
        ss += f * f;
      }
    }
    ss = block_reduce_sum<4>(ss, red);
    float r = rsqrtf(ss / H + eps);
    if (threadIdx.x == 0) inv_rms[row] = r;
    for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
      s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = AttnElem<FP16>::from_f32(AttnElem<FP16>::to_f32(v[j]) * r * AttnElem<FP16>::to_f32(wv[j]));
      *reinterpret_cast<s16x8*>(yr + i) = o;
    }
  }
}

// backward: dx per row + dw accumulated in LDS across this block's rows,
// then one fp32 atomicAdd per element (guide G12: partial-reduce first).
template <bool FP16>
__global__ void rmsnorm_bwd_kernel(const short* __restrict__ dy,
                                   const short* __restrict__ x,
                                   const short* __restrict__ w,
                                   const float* __restrict__ inv_rms,
                                   short* __restrict__ dx,
                                   float* __restrict__ dw,  // fp32 [H]
                                   int rows, int H) {
  extern __shared__ float smem[];  // [H] dw accumulator + 16 reduce
  float* dw_acc = smem;
  float* red = smem + H;
  for (int i = threadIdx.x; i < H; i += blockDim.x) dw_acc[i] = 0.f;
  __syncthreads();
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short* xr = x + (long)row * H;
    const short* dyr = dy + (long)row * H;
    short* dxr = dx + (long)row * H;
    const float r = inv_rms[row];
    // pass 1: ddot = mean(w*dy*xhat)
    float dot = 0.f;
    for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
      s16x8 xv = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 dv = *reinterpret_cast<const s16x8*>(dyr + i);
      s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = AttnElem<FP16>::to_f32(xv[j]) * r;
        dot += AttnElem<FP16>::to_f32(wv[j]) * AttnElem<FP16>::to_f32(dv[j]) * xh;
      }
    }
    dot = block_reduce_sum<4>(dot, red) / H;
    // pass 2: dx = r*(w*dy - xhat*dot); dw_acc += dy*xhat
    for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
      s16x8 xv = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 dv = *reinterpret_cast<const s16x8*>(dyr + i);
      s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = AttnElem<FP16>::to_f32(xv[j]) * r;
        float dyf = AttnElem<FP16>::to_f32(dv[j]);
        o[j] = AttnElem<FP16>::from_f32(r * (AttnElem<FP16>::to_f32(wv[j]) * dyf - xh * dot) +
                           0.f);
        dw_acc[i + j] += dyf * xh;  // each thread owns cols i..i+7
      }
      *reinterpret_cast<s16x8*>(dxr + i) = o;
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < H; i += blockDim.x)
    atomicAdd(&dw[i], dw_acc[i]);
}

// Fast path for H == ITERS*2048 (block 256, 8 elems/thread/iter): the
// add+square pass keeps its values in registers so the normalize pass
// never re-reads the row, and dw-style reductions stay in registers.
// Covers Llama H=4096/8192 (ITERS 2/4); other widths take the generic
// kernels below.
template <int ITERS, bool FP16>
__global__ void add_rmsnorm_fwd_fast(const short* __restrict__ x,
                                     const short* __restrict__ resid_in,
                                     const short* __restrict__ w,
                                     short* __restrict__ y,
                                     short* __restrict__ resid_out,
                                     float* __restrict__ inv_rms, int rows,
                                     float eps, bool has_resid) {
  const int H = ITERS * 2048;
  __shared__ float red[16];
  s16x8 wv[ITERS];
#pragma unroll
  for (int t = 0; t < ITERS; ++t)
    wv[t] = *reinterpret_cast<const s16x8*>(
        w + threadIdx.x * 8 + t * 2048);
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short* xr = x + (long)row * H;
    const short* rr = resid_in + (long)row * H;
    short* yr = y + (long)row * H;
    short* ro = resid_out + (long)row * H;
    float ss = 0.f;
    f32x4 rc[ITERS][2];  // the summed row, fp32, stays in registers
#pragma unroll
    for (int t = 0; t < ITERS; ++t) {
      const int i = threadIdx.x * 8 + t * 2048;
      s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 o;
      if (has_resid) {
        s16x8 rv = *reinterpret_cast<const s16x8*>(rr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = AttnElem<FP16>::to_f32(v[j]) + AttnElem<FP16>::to_f32(rv[j]);
          // match the generic kernel: resid_out stores the bf16 ROUNDING
          // of the sum, and the norm pass reads that rounded value
          o[j] = AttnElem<FP16>::from_f32(f);
          float fr = AttnElem<FP16>::to_f32(o[j]);
          rc[t][j / 4][j % 4] = fr;
          ss += fr * fr;
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = AttnElem<FP16>::to_f32(v[j]);
          o[j] = v[j];
          rc[t][j / 4][j % 4] = f;
          ss += f * f;
        }
      }
      *reinterpret_cast<s16x8*>(ro + i) = o;
    }
    ss = block_reduce_sum<4>(ss, red);
    float r = rsqrtf(ss / H + eps);
    if (threadIdx.x == 0) inv_rms[row] = r;
#pragma unroll
    for (int t = 0; t < ITERS; ++t) {
      const int i = threadIdx.x * 8 + t * 2048;
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = AttnElem<FP16>::from_f32(rc[t][j / 4][j % 4] * r * AttnElem<FP16>::to_f32(wv[t][j]));
      *reinterpret_cast<s16x8*>(yr + i) = o;
    }
  }
}

template <int ITERS, bool FP16>
__global__ void add_rmsnorm_bwd_fast(const short* __restrict__ dy,
                                     const short* __restrict__ dresid,
                                     const short* __restrict__ r_saved,
                                     const short* __restrict__ w,
                                     const float* __restrict__ inv_rms,
                                     short* __restrict__ dx,
                                     float* __restrict__ dw, int rows,
                                     bool has_dresid) {
  const int H = ITERS * 2048;
  __shared__ float red[16];
  s16x8 wv[ITERS];
  f32x4 dwacc[ITERS][2];
#pragma unroll
  for (int t = 0; t < ITERS; ++t) {
    wv[t] = *reinterpret_cast<const s16x8*>(
        w + threadIdx.x * 8 + t * 2048);
    dwacc[t][0] = f32x4{0.f, 0.f, 0.f, 0.f};
    dwacc[t][1] = f32x4{0.f, 0.f, 0.f, 0.f};
  }
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short* xr = r_saved + (long)row * H;
    const short* dyr = dy + (long)row * H;
    const short* drr = dresid + (long)row * H;
    short* dxr = dx + (long)row * H;
    const float rinv = inv_rms[row];
    float dot = 0.f;
    s16x8 xc[ITERS], dc[ITERS];
#pragma unroll
    for (int t = 0; t < ITERS; ++t) {
      const int i = threadIdx.x * 8 + t * 2048;
      xc[t] = *reinterpret_cast<const s16x8*>(xr + i);
      dc[t] = *reinterpret_cast<const s16x8*>(dyr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dot += AttnElem<FP16>::to_f32(wv[t][j]) * AttnElem<FP16>::to_f32(dc[t][j]) *
               (AttnElem<FP16>::to_f32(xc[t][j]) * rinv);
    }
    dot = block_reduce_sum<4>(dot, red) / H;
#pragma unroll
    for (int t = 0; t < ITERS; ++t) {
      const int i = threadIdx.x * 8 + t * 2048;
      s16x8 o;
      s16x8 drv;
      if (has_dresid) drv = *reinterpret_cast<const s16x8*>(drr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = AttnElem<FP16>::to_f32(xc[t][j]) * rinv;
        float dyf = AttnElem<FP16>::to_f32(dc[t][j]);
        float g = rinv * (AttnElem<FP16>::to_f32(wv[t][j]) * dyf - xh * dot);
        if (has_dresid) g += AttnElem<FP16>::to_f32(drv[j]);
        o[j] = AttnElem<FP16>::from_f32(g);
        dwacc[t][j / 4][j % 4] += dyf * xh;
      }
      *reinterpret_cast<s16x8*>(dxr + i) = o;
    }
  }
#pragma unroll
  for (int t = 0; t < ITERS; ++t) {
    const int i = threadIdx.x * 8 + t * 2048;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      atomicAdd(&dw[i + j], dwacc[t][j / 4][j % 4]);
  }
}

// fused residual add + RMSNorm: r = x + resid_in; y = rmsnorm(r) * w.
// Saves one full read+write of the residual stream per call vs separate
// add and norm kernels (and the separate add's backward elementwise).
template <bool FP16>
__global__ void add_rmsnorm_fwd_kernel(const short* __restrict__ x,
                                       const short* __restrict__ resid_in,
                                       const short* __restrict__ w,
                                       short* __restrict__ y,
                                       short* __restrict__ resid_out,
                                       float* __restrict__ inv_rms, int rows,
                                       int H, float eps, bool has_resid) {
  __shared__ float red[16];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short* xr = x + (long)row * H;
    const short* rr = resid_in + (long)row * H;
    short* yr = y + (long)row * H;
    short* ro = resid_out + (long)row * H;
    float ss = 0.f;
    for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
      s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 o;
      if (has_resid) {
        s16x8 rv = *reinterpret_cast<const s16x8*>(rr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = AttnElem<FP16>::to_f32(v[j]) + AttnElem<FP16>::to_f32(rv[j]);
          o[j] = AttnElem<FP16>::from_f32(f);
          ss += f * f;
        }
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = AttnElem<FP16>::to_f32(v[j]);
          o[j] = v[j];
          ss += f * f;
        }
      }
      *reinterpret_cast<s16x8*>(ro + i) = o;
    }
    ss = block_reduce_sum<4>(ss, red);
    float r = rsqrtf(ss / H + eps);
    if (threadIdx.x == 0) inv_rms[row] = r;
    for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
      s16x8 v = *reinterpret_cast<const s16x8*>(ro + i);
      s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = AttnElem<FP16>::from_f32(AttnElem<FP16>::to_f32(v[j]) * r * AttnElem<FP16>::to_f32(wv[j]));
      *reinterpret_cast<s16x8*>(yr + i) = o;
    }
  }
}

// backward: dx = rmsnorm_dx(dy) + dresid (grad into the residual stream
// from downstream); dw accumulated as in rmsnorm_bwd.
template <bool FP16>
__global__ void add_rmsnorm_bwd_kernel(const short* __restrict__ dy,
                                       const short* __restrict__ dresid,
                                       const short* __restrict__ r_saved,
                                       const short* __restrict__ w,
                                       const float* __restrict__ inv_rms,
                                       short* __restrict__ dx,
                                       float* __restrict__ dw, int rows,
                                       int H, bool has_dresid) {
  extern __shared__ float smem[];
  float* dw_acc = smem;
  float* red = smem + H;
  for (int i = threadIdx.x; i < H; i += blockDim.x) dw_acc[i] = 0.f;
  __syncthreads();
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short* xr = r_saved + (long)row * H;
    const short* dyr = dy + (long)row * H;
    const short* drr = dresid + (long)row * H;
    short* dxr = dx + (long)row * H;
    const float rinv = inv_rms[row];
    float dot = 0.f;
    for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
      s16x8 xv = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 dv = *reinterpret_cast<const s16x8*>(dyr + i);
      s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = AttnElem<FP16>::to_f32(xv[j]) * rinv;
        dot += AttnElem<FP16>::to_f32(wv[j]) * AttnElem<FP16>::to_f32(dv[j]) * xh;
      }
    }
    dot = block_reduce_sum<4>(dot, red) / H;
    for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
      s16x8 xv = *reinterpret_cast<const s16x8*>(xr + i);
      s16x8 dv = *reinterpret_cast<const s16x8*>(dyr + i);
      s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = AttnElem<FP16>::to_f32(xv[j]) * rinv;
        float dyf = AttnElem<FP16>::to_f32(dv[j]);
        float g = rinv * (AttnElem<FP16>::to_f32(wv[j]) * dyf - xh * dot);
        if (has_dresid) {
          s16x8 drv = *reinterpret_cast<const s16x8*>(drr + i);
          g += AttnElem<FP16>::to_f32(drv[j]);
        }
        o[j] = AttnElem<FP16>::from_f32(g);
        dw_acc[i + j] += dyf * xh;
      }
      *reinterpret_cast<s16x8*>(dxr + i) = o;
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < H; i += blockDim.x)
    atomicAdd(&dw[i], dw_acc[i]);
}

// fp32 variants (CPU-parity/debug path; also used when model is fp32)
__global__ void rmsnorm_fwd_kernel_f32(const float* __restrict__ x,
                                       const float* __restrict__ w,
                                       float* __restrict__ y,
                                       float* __restrict__ inv_rms, int rows,
                                       int H, float eps) {
  __shared__ float red[16];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const float* xr = x + (long)row * H;
    float* yr = y + (long)row * H;
    float ss = 0.f;
    for (int i = threadIdx.x; i < H; i += blockDim.x) ss += xr[i] * xr[i];
    ss = block_reduce_sum<4>(ss, red);
    float r = rsqrtf(ss / H + eps);
    if (threadIdx.x == 0) inv_rms[row] = r;
    for (int i = threadIdx.x; i < H; i += blockDim.x)
      yr[i] = xr[i] * r * w[i];
  }
}

// ---------------------------------------------------------------------------
// RoPE (fused q + k, neox half-split)
// ---------------------------------------------------------------------------
// x [b, s, h, d]; cos/sin [s, d/2] fp32. Each thread: 8 first-half elements
// + 8 matching second-half elements of one (b,s,h) row.

template <bool FP16>
__global__ void rope_kernel(const short* __restrict__ x,
                            short* __restrict__ out,
                            const float* __restrict__ cosp,
                            const float* __restrict__ sinp, long total_rows,
                            int H, int D, int S) {
  const int d2 = D / 2;
  const int chunks = d2 / 8;  // d2 multiple of 8
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long n = total_rows * chunks;
  for (; idx < n; idx += (long)gridDim.x * blockDim.x) {
    const long rowhead = idx / chunks;   // (b*s*h)
    const int c = idx % chunks;
    const int s = (rowhead / H) % S;
    const short* xr = x + rowhead * D + c * 8;
    short* orow = out + rowhead * D + c * 8;
    const float* cp = cosp + (long)s * d2 + c * 8;
    const float* sp = sinp + (long)s * d2 + c * 8;
    s16x8 x1 = *reinterpret_cast<const s16x8*>(xr);
    s16x8 x2 = *reinterpret_cast<const s16x8*>(xr + d2);
    f32x4 c0 = *reinterpret_cast<const f32x4*>(cp);
    f32x4 c1 = *reinterpret_cast<const f32x4*>(cp + 4);
    f32x4 s0 = *reinterpret_cast<const f32x4*>(sp);
    f32x4 s1 = *reinterpret_cast<const f32x4*>(sp + 4);
    s16x8 o1, o2;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float cj = j < 4 ? c0[j] : c1[j - 4];
      float sj = j < 4 ? s0[j] : s1[j - 4];
      float a = AttnElem<FP16>::to_f32(x1[j]);
      float b = AttnElem<FP16>::to_f32(x2[j]);
      o1[j] = AttnElem<FP16>::from_f32(a * cj - b * sj);
      o2[j] = AttnElem<FP16>::from_f32(b * cj + a * sj);
    }
    *reinterpret_cast<s16x8*>(orow) = o1;
    *reinterpret_cast<s16x8*>(orow + d2) = o2;
  }
}

// ---------------------------------------------------------------------------
// SwiGLU
// ---------------------------------------------------------------------------


// 2x-wide variants: two s16x8 (32 B) per thread per trip doubles the
// loads in flight per wave — the 1x versions measured ~60% of HBM peak.
template <bool FP16>
__global__ void swiglu_fwd_kernel_v2(const short* __restrict__ g,
                                     const short* __restrict__ u,
                                     short* __restrict__ y, long n16) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; idx < n16; idx += (long)gridDim.x * blockDim.x) {
    const long base = idx * 16;
    s16x8 gv0 = *reinterpret_cast<const s16x8*>(g + base);
    s16x8 gv1 = *reinterpret_cast<const s16x8*>(g + base + 8);
    s16x8 uv0 = *reinterpret_cast<const s16x8*>(u + base);
    s16x8 uv1 = *reinterpret_cast<const s16x8*>(u + base + 8);
    s16x8 o0, o1;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf0 = AttnElem<FP16>::to_f32(gv0[j]);
      float gf1 = AttnElem<FP16>::to_f32(gv1[j]);
      float s0 = 1.f / (1.f + __expf(-gf0));
      float s1 = 1.f / (1.f + __expf(-gf1));
      o0[j] = AttnElem<FP16>::from_f32(gf0 * s0 * AttnElem<FP16>::to_f32(uv0[j]));
      o1[j] = AttnElem<FP16>::from_f32(gf1 * s1 * AttnElem<FP16>::to_f32(uv1[j]));
    }
    *reinterpret_cast<s16x8*>(y + base) = o0;
    *reinterpret_cast<s16x8*>(y + base + 8) = o1;
  }
}

template <bool FP16>
__global__ void swiglu_bwd_kernel_v2(const short* __restrict__ dy,
                                     const short* __restrict__ g,
                                     const short* __restrict__ u,
                                     short* __restrict__ dg,
                                     short* __restrict__ du, long n16) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; idx < n16; idx += (long)gridDim.x * blockDim.x) {
    const long base = idx * 16;
    s16x8 dv0 = *reinterpret_cast<const s16x8*>(dy + base);
    s16x8 dv1 = *reinterpret_cast<const s16x8*>(dy + base + 8);
    s16x8 gv0 = *reinterpret_cast<const s16x8*>(g + base);
    s16x8 gv1 = *reinterpret_cast<const s16x8*>(g + base + 8);
    s16x8 uv0 = *reinterpret_cast<const s16x8*>(u + base);
    s16x8 uv1 = *reinterpret_cast<const s16x8*>(u + base + 8);
    s16x8 og0, og1, ou0, ou1;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf0 = AttnElem<FP16>::to_f32(gv0[j]);
      float gf1 = AttnElem<FP16>::to_f32(gv1[j]);
      float d0 = AttnElem<FP16>::to_f32(dv0[j]);
      float d1 = AttnElem<FP16>::to_f32(dv1[j]);
      float s0 = 1.f / (1.f + __expf(-gf0));
      float s1 = 1.f / (1.f + __expf(-gf1));
      og0[j] = AttnElem<FP16>::from_f32(d0 * AttnElem<FP16>::to_f32(uv0[j]) *
                           (s0 * (1.f + gf0 * (1.f - s0))));
      og1[j] = AttnElem<FP16>::from_f32(d1 * AttnElem<FP16>::to_f32(uv1[j]) *
                           (s1 * (1.f + gf1 * (1.f - s1))));
      ou0[j] = AttnElem<FP16>::from_f32(d0 * gf0 * s0);
      ou1[j] = AttnElem<FP16>::from_f32(d1 * gf1 * s1);
    }
    *reinterpret_cast<s16x8*>(dg + base) = og0;
    *reinterpret_cast<s16x8*>(dg + base + 8) = og1;
    *reinterpret_cast<s16x8*>(du + base) = ou0;
    *reinterpret_cast<s16x8*>(du + base + 8) = ou1;
  }
}

template <bool FP16>
__global__ void swiglu_fwd_kernel(const short* __restrict__ g,
                                  const short* __restrict__ u,
                                  short* __restrict__ y, long n8) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; idx < n8; idx += (long)gridDim.x * blockDim.x) {
    s16x8 gv = *reinterpret_cast<const s16x8*>(g + idx * 8);
    s16x8 uv = *reinterpret_cast<const s16x8*>(u + idx * 8);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = AttnElem<FP16>::to_f32(gv[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      o[j] = AttnElem<FP16>::from_f32(gf * sig * AttnElem<FP16>::to_f32(uv[j]));
    }
    *reinterpret_cast<s16x8*>(y + idx * 8) = o;
  }
}

template <bool FP16>
__global__ void swiglu_bwd_kernel(const short* __restrict__ dy,
                                  const short* __restrict__ g,
                                  const short* __restrict__ u,
                                  short* __restrict__ dg,
                                  short* __restrict__ du, long n8) {
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; idx < n8; idx += (long)gridDim.x * blockDim.x) {
    s16x8 dv = *reinterpret_cast<const s16x8*>(dy + idx * 8);
    s16x8 gv = *reinterpret_cast<const s16x8*>(g + idx * 8);
    s16x8 uv = *reinterpret_cast<const s16x8*>(u + idx * 8);
    s16x8 og, ou;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = AttnElem<FP16>::to_f32(gv[j]);
      float uf = AttnElem<FP16>::to_f32(uv[j]);
      float d = AttnElem<FP16>::to_f32(dv[j]);
      float sig = 1.f / (1.f + __expf(-gf));
      float silu = gf * sig;
      og[j] = AttnElem<FP16>::from_f32(d * uf * (sig * (1.f + gf * (1.f - sig))));
      ou[j] = AttnElem<FP16>::from_f32(d * silu);
    }
    *reinterpret_cast<s16x8*>(dg + idx * 8) = og;
    *reinterpret_cast<s16x8*>(du + idx * 8) = ou;
  }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

static inline int grid_for(long work, int block) {
  long g = (work + block - 1) / block;
  return (int)std::min<long>(g, 2048);
}

std::vector<torch::Tensor> rmsnorm_forward(torch::Tensor x, torch::Tensor w,
                                           double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int H = x.size(-1);
  const long rows = x.numel() / H;
  auto y = torch::empty_like(x);
  auto inv_rms = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = grid_for(rows, 1);
  if (x.scalar_type() == torch::kBFloat16 ||
      x.scalar_type() == torch::kHalf) {
    TORCH_CHECK(H % 8 == 0, "hidden size must be a multiple of 8");
    FP16_SWITCH(x.scalar_type() == torch::kHalf,
        hipLaunchKernelGGL((rmsnorm_fwd_kernel<kFP16>),
                           dim3(std::min<long>(rows, 2048)),
                           dim3(256), 0, stream,
                           (const short*)x.data_ptr(),
                           (const short*)w.data_ptr(),
                           (short*)y.data_ptr(), inv_rms.data_ptr<float>(),
                           (int)rows, H, (float)eps));
  } else if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(rmsnorm_fwd_kernel_f32,
                       dim3(std::min<long>(rows, 2048)), dim3(256), 0, stream,
                       x.data_ptr<float>(), w.data_ptr<float>(),
                       y.data_ptr<float>(), inv_rms.data_ptr<float>(),
                       (int)rows, H, (float)eps);
  } else {
    TORCH_CHECK(false, "rmsnorm: unsupported dtype");
  }
  HIP_CHECK_LAST();
  return {y, inv_rms};
}

std::vector<torch::Tensor> rmsnorm_backward(torch::Tensor dy, torch::Tensor x,
                                            torch::Tensor w,
                                            torch::Tensor inv_rms) {
  TORCH_CHECK(x.is_cuda() && (x.scalar_type() == torch::kBFloat16 ||
                              x.scalar_type() == torch::kHalf),
              "rmsnorm backward: bf16/fp16 only on GPU");
  const bool f16 = x.scalar_type() == torch::kHalf;
  const int H = x.size(-1);
  const long rows = x.numel() / H;
  auto dx = torch::empty_like(x);
  auto dw32 = torch::zeros({H}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  const int lds = (H + 16) * sizeof(float);
  FP16_SWITCH(f16,
      hipLaunchKernelGGL((rmsnorm_bwd_kernel<kFP16>),
                         dim3(std::min<long>(rows, 512)),
                         dim3(256), lds, stream,
                         (const short*)dy.data_ptr(),
                         (const short*)x.data_ptr(),
                         (const short*)w.data_ptr(),
                         inv_rms.data_ptr<float>(), (short*)dx.data_ptr(),
                         dw32.data_ptr<float>(), (int)rows, H));
  HIP_CHECK_LAST();
  return {dx, dw32.to(w.scalar_type())};
}

std::vector<torch::Tensor> rope_forward(torch::Tensor q, torch::Tensor k,
                                        torch::Tensor cos, torch::Tensor sin) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous());
  TORCH_CHECK(q.scalar_type() == torch::kBFloat16 ||
                  q.scalar_type() == torch::kHalf,
              "rope: bf16/fp16 only on GPU");
  const bool f16 = q.scalar_type() == torch::kHalf;
  TORCH_CHECK(cos.scalar_type() == torch::kFloat32);
  TORCH_CHECK(cos.is_cuda() && sin.is_cuda(),
              "rope: cos/sin tables must be on the GPU");
  const int D = q.size(-1);
  TORCH_CHECK(D % 16 == 0, "rope: head_dim must be a multiple of 16");
  const int S = q.size(1);
  auto cosc = cos.contiguous();
  auto sinc = sin.contiguous();
  auto qo = torch::empty_like(q);
  auto ko = torch::empty_like(k);
  auto stream = at::hip::getCurrentHIPStream();
  {
    const long rows = q.numel() / D;
    const long work = rows * (D / 16);
    FP16_SWITCH(f16,
        hipLaunchKernelGGL((rope_kernel<kFP16>), dim3(grid_for(work, 256)),
                           dim3(256), 0, stream,
                           (const short*)q.data_ptr(),
                           (short*)qo.data_ptr(), cosc.data_ptr<float>(),
                           sinc.data_ptr<float>(), rows, (int)q.size(2), D,
                           S));
  }
  {
    const long rows = k.numel() / D;
    const long work = rows * (D / 16);
    FP16_SWITCH(f16,
        hipLaunchKernelGGL((rope_kernel<kFP16>), dim3(grid_for(work, 256)),
                           dim3(256), 0, stream,
                           (const short*)k.data_ptr(),
                           (short*)ko.data_ptr(), cosc.data_ptr<float>(),
                           sinc.data_ptr<float>(), rows, (int)k.size(2), D,
                           S));
  }
  HIP_CHECK_LAST();
  return {qo, ko};
}

torch::Tensor swiglu_forward(torch::Tensor g, torch::Tensor u) {
  TORCH_CHECK(g.is_cuda() && (g.scalar_type() == torch::kBFloat16 ||
                              g.scalar_type() == torch::kHalf),
              "swiglu: bf16/fp16 only on GPU");
  const bool f16 = g.scalar_type() == torch::kHalf;
  TORCH_CHECK(g.numel() % 8 == 0);
  auto y = torch::empty_like(g);
  auto stream = at::hip::getCurrentHIPStream();
  const long n8 = g.numel() / 8;
  if (g.numel() % 16 == 0) {
    const long n16 = g.numel() / 16;
    FP16_SWITCH(f16,
        hipLaunchKernelGGL((swiglu_fwd_kernel_v2<kFP16>),
                           dim3(grid_for(n16, 256)), dim3(256), 0, stream,
                           (const short*)g.data_ptr(),
                           (const short*)u.data_ptr(),
                           (short*)y.data_ptr(), n16));
  } else {
    FP16_SWITCH(f16,
        hipLaunchKernelGGL((swiglu_fwd_kernel<kFP16>),
                           dim3(grid_for(n8, 256)), dim3(256), 0, stream,
                           (const short*)g.data_ptr(),
                           (const short*)u.data_ptr(),
                           (short*)y.data_ptr(), n8));
  }
  HIP_CHECK_LAST();
  return y;
}

std::vector<torch::Tensor> swiglu_backward(torch::Tensor dy, torch::Tensor g,
                                           torch::Tensor u) {
  TORCH_CHECK(g.is_cuda() && (g.scalar_type() == torch::kBFloat16 ||
                              g.scalar_type() == torch::kHalf));
  const bool f16 = g.scalar_type() == torch::kHalf;
  auto dg = torch::empty_like(g);
  auto du = torch::empty_like(u);
  auto stream = at::hip::getCurrentHIPStream();
  const long n8 = g.numel() / 8;
  if (dy.numel() % 16 == 0) {
    const long n16 = dy.numel() / 16;
    FP16_SWITCH(f16,
        hipLaunchKernelGGL((swiglu_bwd_kernel_v2<kFP16>),
                           dim3(grid_for(n16, 256)), dim3(256), 0, stream,
                           (const short*)dy.data_ptr(),
                           (const short*)g.data_ptr(),
                           (const short*)u.data_ptr(),
                           (short*)dg.data_ptr(), (short*)du.data_ptr(),
                           n16));
  } else {
    FP16_SWITCH(f16,
        hipLaunchKernelGGL((swiglu_bwd_kernel<kFP16>),
                           dim3(grid_for(n8, 256)), dim3(256), 0, stream,
                           (const short*)dy.data_ptr(),
                           (const short*)g.data_ptr(),
                           (const short*)u.data_ptr(),
                           (short*)dg.data_ptr(), (short*)du.data_ptr(),
                           n8));
  }
  HIP_CHECK_LAST();
  return {dg, du};
}


std::vector<torch::Tensor> add_rmsnorm_forward(torch::Tensor x,
                                               torch::Tensor resid,
                                               torch::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 ||
                  x.scalar_type() == torch::kHalf,
              "add_rmsnorm: bf16/fp16 only on GPU");
  const bool f16 = x.scalar_type() == torch::kHalf;
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0);
  const long rows = x.numel() / H;
  const bool has_resid = resid.numel() > 0;
  auto y = torch::empty_like(x);
  auto resid_out = torch::empty_like(x);
  auto inv_rms = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  const short* xin = (const short*)x.data_ptr();
  const short* rin = has_resid ? (const short*)resid.data_ptr() : xin;
  const short* wp = (const short*)w.data_ptr();
  short* yp = (short*)y.data_ptr();
  short* rop = (short*)resid_out.data_ptr();
  float* irp = inv_rms.data_ptr<float>();
  dim3 grid(std::min<long>(rows, 2048)), block(256);
  FP16_SWITCH(f16,
      if (H == 4096) {
        hipLaunchKernelGGL((add_rmsnorm_fwd_fast<2, kFP16>), grid, block, 0,
                           stream, xin, rin, wp, yp, rop, irp, (int)rows,
                           (float)eps, has_resid);
      } else if (H == 8192) {
        hipLaunchKernelGGL((add_rmsnorm_fwd_fast<4, kFP16>), grid, block, 0,
                           stream, xin, rin, wp, yp, rop, irp, (int)rows,
                           (float)eps, has_resid);
      } else if (H == 2048) {
        hipLaunchKernelGGL((add_rmsnorm_fwd_fast<1, kFP16>), grid, block, 0,
                           stream, xin, rin, wp, yp, rop, irp, (int)rows,
                           (float)eps, has_resid);
      } else {
        hipLaunchKernelGGL((add_rmsnorm_fwd_kernel<kFP16>), grid, block, 0,
                           stream, xin, rin, wp, yp, rop, irp, (int)rows, H,
                           (float)eps, has_resid);
      });
  HIP_CHECK_LAST();
  return {y, resid_out, inv_rms};
}

std::vector<torch::Tensor> add_rmsnorm_backward(torch::Tensor dy,
                                                torch::Tensor dresid,
                                                torch::Tensor r_saved,
                                                torch::Tensor w,
                                                torch::Tensor inv_rms) {
  const int H = r_saved.size(-1);
  const long rows = r_saved.numel() / H;
  const bool has_dresid = dresid.numel() > 0;
  auto dx = torch::empty_like(r_saved);
  auto dw32 = torch::zeros({H}, r_saved.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  const short* dyp = (const short*)dy.data_ptr();
  const short* drp = has_dresid ? (const short*)dresid.data_ptr() : dyp;
  const short* xp = (const short*)r_saved.data_ptr();
  const short* wp = (const short*)w.data_ptr();
  const float* irp = inv_rms.data_ptr<float>();
  short* dxp = (short*)dx.data_ptr();
  float* dwp = dw32.data_ptr<float>();
  dim3 block(256);
  const bool f16 = r_saved.scalar_type() == torch::kHalf;
  FP16_SWITCH(f16,
      if (H == 4096) {
        hipLaunchKernelGGL((add_rmsnorm_bwd_fast<2, kFP16>),
                           dim3(std::min<long>(rows, 512)), block, 0,
                           stream, dyp, drp, xp, wp, irp, dxp, dwp,
                           (int)rows, has_dresid);
      } else if (H == 8192) {
        hipLaunchKernelGGL((add_rmsnorm_bwd_fast<4, kFP16>),
                           dim3(std::min<long>(rows, 512)), block, 0,
                           stream, dyp, drp, xp, wp, irp, dxp, dwp,
                           (int)rows, has_dresid);
      } else if (H == 2048) {
        hipLaunchKernelGGL((add_rmsnorm_bwd_fast<1, kFP16>),
                           dim3(std::min<long>(rows, 512)), block, 0,
                           stream, dyp, drp, xp, wp, irp, dxp, dwp,
                           (int)rows, has_dresid);
      } else {
        const int lds = (H + 16) * sizeof(float);
        hipLaunchKernelGGL((add_rmsnorm_bwd_kernel<kFP16>),
                           dim3(std::min<long>(rows, 512)), block, lds,
                           stream, dyp, drp, xp, wp, irp, dxp, dwp,
                           (int)rows, H, has_dresid);
      });
  HIP_CHECK_LAST();
  return {dx, dw32.to(w.scalar_type())};
}
