// Common CDNA4 (gfx950) kernel utilities.
//
// Design notes (per /opt/skills guides):
//  - wave = 64 lanes; block sizes are multiples of 64
//  - bf16 memory ops vectorized as short4/short8 reinterprets (hipcc does
//    not auto-vectorize bf16 scalar loads)
//  - memory-bound grids capped ~2048 blocks with grid-stride loops
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define DEVINLINE __device__ __forceinline__

typedef __hip_bfloat16 bf16_t;

// vector types for wide loads
typedef short  s16x4 __attribute__((ext_vector_type(4)));
typedef short  s16x8 __attribute__((ext_vector_type(8)));
typedef float  f32x2 __attribute__((ext_vector_type(2)));
typedef float  f32x4 __attribute__((ext_vector_type(4)));
typedef float  f32x16 __attribute__((ext_vector_type(16)));
typedef short  bf16x8 __attribute__((ext_vector_type(8)));   // MFMA A/B frag
typedef unsigned int u32x2 __attribute__((ext_vector_type(2)));

DEVINLINE float bf16_to_f32(short u) {
  union { float f; unsigned int i; } cv;
  cv.i = ((unsigned int)(unsigned short)u) << 16;
  return cv.f;
}

DEVINLINE short f32_to_bf16(float f) {
  // round-to-nearest-even
  union { float f; unsigned int i; } cv;
  cv.f = f;
  unsigned int x = cv.i;
  unsigned int rounded = x + 0x7fff + ((x >> 16) & 1);
  if ((x & 0x7f800000u) == 0x7f800000u) rounded = x;  // inf/nan passthrough
  return (short)(rounded >> 16);
}

// wave-wide reductions (64 lanes)
DEVINLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

DEVINLINE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// block reduction via LDS (block = N waves, N <= 16)
template <int NWAVES>
DEVINLINE float block_reduce_sum(float v, float* lds_scratch) {
  v = wave_reduce_sum(v);
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  float r = (lane < NWAVES) ? lds_scratch[lane] : 0.f;
  r = wave_reduce_sum(r);  // all lanes end with the total
  __syncthreads();
  return r;
}

#define HIP_CHECK_LAST()                                                    \
  do {                                                                      \
    hipError_t e_ = hipGetLastError();                                      \
    if (e_ != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP kernel launch failed: ",                      \
                  hipGetErrorString(e_));                                   \
    }                                                                       \
  } while (0)
