// Fused cross-entropy for gfx950.
//
// forward: one 256-thread block per row chunk; online max+sum in one pass
// over the vocab (memory-bound: bf16x8 loads), emits per-row lse and the
// summed loss over valid rows.
// backward: dlogits = (softmax - onehot) * scale in one vectorized pass.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"
#include "attn_common.h"

template <bool FP16>
__global__ void ce_fwd_kernel(const short* __restrict__ logits,
                              const long* __restrict__ target,
                              float* __restrict__ lse,
                              float* __restrict__ loss_sum,
                              int* __restrict__ nvalid, int rows, int V,
                              long ignore_index) {
  __shared__ float red[16];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const short* lr = logits + (long)row * V;
    // online max & sum
    float m = -INFINITY, s = 0.f;
    for (int i = threadIdx.x * 8; i < V; i += blockDim.x * 8) {
      float vals[8];
      if (i + 8 <= V) {
        s16x8 v = *reinterpret_cast<const s16x8*>(lr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) vals[j] = AttnElem<FP16>::to_f32(v[j]);
      } else {
        for (int j = 0; j < 8; ++j)
          vals[j] = (i + j < V) ? AttnElem<FP16>::to_f32(lr[i + j]) : -INFINITY;
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float x = vals[j];
        if (x > m) {
          s *= __expf(m - x);
          m = x;
        }
        s += __expf(x - m);
      }
    }
    // block merge of (m, s)
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float mo = __shfl_xor(m, off, 64);
      float so = __shfl_xor(s, off, 64);
      float mn = fmaxf(m, mo);
      // guard -inf slices (threads whose vocab slice was empty):
      // exp(-inf - -inf) would be NaN
      float wa = (m == -INFINITY) ? 0.f : __expf(m - mn);
      float wb = (mo == -INFINITY) ? 0.f : __expf(mo - mn);
      s = s * wa + so * wb;
      m = mn;
    }
    __shared__ float mred[8], sred[8];
    if (lane == 0) { mred[wid] = m; sred[wid] = s; }
    __syncthreads();
    if (threadIdx.x == 0) {
      float mm = mred[0], ss2 = sred[0];
      for (int wv = 1; wv < blockDim.x / WAVE; ++wv) {
        float mn = fmaxf(mm, mred[wv]);
        float wa = (mm == -INFINITY) ? 0.f : __expf(mm - mn);
        float wb = (mred[wv] == -INFINITY) ? 0.f : __expf(mred[wv] - mn);
        ss2 = ss2 * wa + sred[wv] * wb;
        mm = mn;
      }
      float l = mm + __logf(ss2);
      lse[row] = l;
      long t = target[row];
      if (t != ignore_index) {
        float picked = AttnElem<FP16>::to_f32(lr[t]);
        atomicAdd(loss_sum, l - picked);
        atomicAdd(nvalid, 1);
      }
    }
    __syncthreads();
  }
  (void)red;
}

template <bool FP16>
__global__ void ce_bwd_kernel(const short* __restrict__ logits,
                              const long* __restrict__ target,
                              const float* __restrict__ lse,
                              const float* __restrict__ scale,
                              short* __restrict__ dlogits, int rows, int V,
                              long ignore_index) {
  const float sc = *scale;
  long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long n = (long)rows * ((V + 7) / 8);
  const int chunks = (V + 7) / 8;
  for (; idx < n; idx += (long)gridDim.x * blockDim.x) {
    const int row = idx / chunks;
    const int i = (idx % chunks) * 8;
    const long t = target[row];
    const bool valid = (t != ignore_index);
    const float l = lse[row];
    const short* lr = logits + (long)row * V;
    short* dr = dlogits + (long)row * V;
    if (i + 8 <= V) {
      s16x8 v = *reinterpret_cast<const s16x8*>(lr + i);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float soft = __expf(AttnElem<FP16>::to_f32(v[j]) - l);
        if (valid && (long)(i + j) == t) soft -= 1.f;
        o[j] = AttnElem<FP16>::from_f32(valid ? soft * sc : 0.f);
      }
      *reinterpret_cast<s16x8*>(dr + i) = o;
    } else {
      for (int j = 0; i + j < V; ++j) {
        float soft = __expf(AttnElem<FP16>::to_f32(lr[i + j]) - l);
        if (valid && (long)(i + j) == t) soft -= 1.f;
        dr[i + j] = AttnElem<FP16>::from_f32(valid ? soft * sc : 0.f);
      }
    }
  }
}

std::vector<torch::Tensor> cross_entropy_forward(torch::Tensor logits,
                                                 torch::Tensor target,
                                                 long ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2);
  const bool f16 = logits.scalar_type() == torch::kHalf;
  TORCH_CHECK(logits.scalar_type() == torch::kBFloat16 || f16,
              "cross_entropy: bf16 logits only on GPU");
  TORCH_CHECK(target.scalar_type() == torch::kLong);
  const int rows = logits.size(0), V = logits.size(1);
  auto lse = torch::empty({rows}, logits.options().dtype(torch::kFloat32));
  auto loss = torch::zeros({}, logits.options().dtype(torch::kFloat32));
  auto nvalid = torch::zeros({}, logits.options().dtype(torch::kInt32));
  auto stream = at::hip::getCurrentHIPStream();
  FP16_SWITCH(f16,
      hipLaunchKernelGGL((ce_fwd_kernel<kFP16>),
                         dim3(std::min(rows, 2048)), dim3(256), 0, stream,
                         (const short*)logits.data_ptr(),
                         target.data_ptr<long>(), lse.data_ptr<float>(),
                         loss.data_ptr<float>(), nvalid.data_ptr<int>(),
                         rows, V, ignore_index));
  HIP_CHECK_LAST();
  return {loss, nvalid, lse};
}

torch::Tensor cross_entropy_backward(torch::Tensor logits,
                                     torch::Tensor target, torch::Tensor lse,
                                     torch::Tensor scale, long ignore_index) {
  const int rows = logits.size(0), V = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  auto stream = at::hip::getCurrentHIPStream();
  auto scale_f = scale.to(torch::kFloat32);
  const long work = (long)rows * ((V + 7) / 8);
  int grid = (int)std::min<long>((work + 255) / 256, 2048);
  const bool f16b = logits.scalar_type() == torch::kHalf;
  FP16_SWITCH(f16b,
      hipLaunchKernelGGL((ce_bwd_kernel<kFP16>), dim3(grid), dim3(256), 0,
                         stream, (const short*)logits.data_ptr(),
                         target.data_ptr<long>(), lse.data_ptr<float>(),
                         scale_f.data_ptr<float>(),
                         (short*)dlogits.data_ptr(), rows, V,
                         ignore_index));
  HIP_CHECK_LAST();
  return dlogits;
}
