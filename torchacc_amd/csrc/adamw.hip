// Fused AdamW for gfx950 with syncfree found_inf gating.
//
// One grid-stride kernel per (large, flat) parameter tensor — the FSDP
// engine presents few multi-hundred-MB flat shards, so per-tensor launch
// overhead is negligible and the kernel is purely HBM-bound. fp32
// exp_avg/exp_avg_sq (+ optional fp32 master weights for bf16 params); the
// device-side found_inf flag makes the whole update a no-op without any
// host synchronization (reference: torch_xla syncfree optimizers,
// utils/patch.py:55-57).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

// HALF=false: bf16 params/grads, HALF=true: fp16
template <bool MASTER, bool HALF = false>
__global__ void adamw_bf16_kernel(short* __restrict__ p,
                                  const short* __restrict__ g,
                                  float* __restrict__ m,
                                  float* __restrict__ v,
                                  float* __restrict__ master,
                                  const float* __restrict__ found_inf,
                                  long n, float lr, float beta1, float beta2,
                                  float eps, float wd, float bc1, float bc2) {
  if (*found_inf != 0.f) return;
  auto ld = [](short u) {
    if (HALF) {
      __half h = *reinterpret_cast<const __half*>(&u);
      return __half2float(h);
    }
    return bf16_to_f32(u);
  };
  auto st = [](float f) {
    if (HALF) {
      __half h = __float2half(f);
      return *reinterpret_cast<short*>(&h);
    }
    return f32_to_bf16(f);
  };
  long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const long stride = (long)gridDim.x * blockDim.x * 4;
  for (; i + 4 <= n; i += stride) {
    s16x4 gv = *reinterpret_cast<const s16x4*>(g + i);
    f32x4 mv = *reinterpret_cast<const f32x4*>(m + i);
    f32x4 vv = *reinterpret_cast<const f32x4*>(v + i);
    f32x4 pv;
    if (MASTER) {
      pv = *reinterpret_cast<const f32x4*>(master + i);
    } else {
      s16x4 pb = *reinterpret_cast<const s16x4*>(p + i);
#pragma unroll
      for (int j = 0; j < 4; ++j) pv[j] = ld(pb[j]);
    }
    s16x4 po;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gf = ld(gv[j]);
      mv[j] = beta1 * mv[j] + (1.f - beta1) * gf;
      vv[j] = beta2 * vv[j] + (1.f - beta2) * gf * gf;
      float denom = sqrtf(vv[j] / bc2) + eps;
      float upd = (mv[j] / bc1) / denom;
      pv[j] = pv[j] * (1.f - lr * wd) - lr * upd;
      po[j] = st(pv[j]);
    }
    *reinterpret_cast<f32x4*>(m + i) = mv;
    *reinterpret_cast<f32x4*>(v + i) = vv;
    if (MASTER) *reinterpret_cast<f32x4*>(master + i) = pv;
    *reinterpret_cast<s16x4*>(p + i) = po;
  }
  // tail (n not multiple of 4)
  if (i < n && i + 4 > n) {
    for (; i < n; ++i) {
      float gf = ld(g[i]);
      float mi = beta1 * m[i] + (1.f - beta1) * gf;
      float vi = beta2 * v[i] + (1.f - beta2) * gf * gf;
      float pvi = MASTER ? master[i] : ld(p[i]);
      pvi = pvi * (1.f - lr * wd) - lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
      m[i] = mi;
      v[i] = vi;
      if (MASTER) master[i] = pvi;
      p[i] = st(pvi);
    }
  }
}

__global__ void adamw_f32_kernel(float* __restrict__ p,
                                 const float* __restrict__ g,
                                 float* __restrict__ m,
                                 float* __restrict__ v,
                                 const float* __restrict__ found_inf, long n,
                                 float lr, float beta1, float beta2,
                                 float eps, float wd, float bc1, float bc2) {
  if (*found_inf != 0.f) return;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (long)gridDim.x * blockDim.x) {
    float gf = g[i];
    float mi = beta1 * m[i] + (1.f - beta1) * gf;
    float vi = beta2 * v[i] + (1.f - beta2) * gf * gf;
    float pv = p[i] * (1.f - lr * wd) -
               lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
    m[i] = mi;
    v[i] = vi;
    p[i] = pv;
  }
}

void fused_adamw(std::vector<torch::Tensor> params,
                 std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> exp_avgs,
                 std::vector<torch::Tensor> exp_avg_sqs,
                 std::vector<torch::Tensor> masters,
                 std::vector<double> steps, torch::Tensor found_inf,
                 double lr, double beta1, double beta2, double eps,
                 double wd) {
  auto stream = at::hip::getCurrentHIPStream();
  for (size_t t = 0; t < params.size(); ++t) {
    auto& p = params[t];
    const long n = p.numel();
    const float bc1 = 1.f - powf((float)beta1, (float)steps[t]);
    const float bc2 = 1.f - powf((float)beta2, (float)steps[t]);
    const int grid = (int)std::min<long>((n + 1023) / 1024, 2048);
    if (p.scalar_type() == torch::kBFloat16 ||
        p.scalar_type() == torch::kHalf) {
      const bool has_master = masters[t].numel() > 0;
      const bool half = p.scalar_type() == torch::kHalf;
#define ADAMW_ARGS                                                           \
  (short*)p.data_ptr(), (const short*)grads[t].data_ptr(),                   \
      exp_avgs[t].data_ptr<float>(), exp_avg_sqs[t].data_ptr<float>(),       \
      has_master ? masters[t].data_ptr<float>() : (float*)nullptr,           \
      found_inf.data_ptr<float>(), n, (float)lr, (float)beta1,               \
      (float)beta2, (float)eps, (float)wd, bc1, bc2
      if (has_master) {
        if (half)
          hipLaunchKernelGGL((adamw_bf16_kernel<true, true>), dim3(grid),
                             dim3(256), 0, stream, ADAMW_ARGS);
        else
          hipLaunchKernelGGL((adamw_bf16_kernel<true, false>), dim3(grid),
                             dim3(256), 0, stream, ADAMW_ARGS);
      } else {
        if (half)
          hipLaunchKernelGGL((adamw_bf16_kernel<false, true>), dim3(grid),
                             dim3(256), 0, stream, ADAMW_ARGS);
        else
          hipLaunchKernelGGL((adamw_bf16_kernel<false, false>), dim3(grid),
                             dim3(256), 0, stream, ADAMW_ARGS);
      }
#undef ADAMW_ARGS
    } else if (p.scalar_type() == torch::kFloat32) {
      hipLaunchKernelGGL(adamw_f32_kernel, dim3(grid), dim3(256), 0, stream,
                         p.data_ptr<float>(), grads[t].data_ptr<float>(),
                         exp_avgs[t].data_ptr<float>(),
                         exp_avg_sqs[t].data_ptr<float>(),
                         found_inf.data_ptr<float>(), n, (float)lr,
                         (float)beta1, (float)beta2, (float)eps, (float)wd,
                         bc1, bc2);
    } else {
      TORCH_CHECK(false, "fused_adamw: unsupported dtype");
    }
  }
  HIP_CHECK_LAST();
}
