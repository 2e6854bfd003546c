// Fused Adam step (reference Optimizers.cu:145 AdamCuda): m/v update, bias
// correction, weight decay, fp32 master update and optional bf16/fp16
// re-materialization — one pass over the parameter.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {
constexpr int BLOCK = 256;

template <typename G, typename P16>
__global__ void adam_kernel(float* __restrict__ p, const G* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            P16* __restrict__ out16, int64_t n, float lr,
                            float b1, float b2, float eps, float wd,
                            float bc1_, float bc2_,
                            const float* __restrict__ bc_dev) {
  const float bc1 = bc_dev ? bc_dev[0] : bc1_;
  const float bc2 = bc_dev ? bc_dev[1] : bc2_;
  int64_t i = ((int64_t)blockIdx.x * BLOCK + threadIdx.x) * 4;
  const int64_t stride = (int64_t)gridDim.x * BLOCK * 4;
  for (; i < n; i += stride) {
    if (i + 4 <= n) {
      float4v pv = *reinterpret_cast<float4v*>(p + i);
      float4v mv = *reinterpret_cast<float4v*>(m + i);
      float4v vv = *reinterpret_cast<float4v*>(v + i);
      float gg[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) gg[j] = (float)g[i + j];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float gf = gg[j] + wd * pv.v[j];
        mv.v[j] = b1 * mv.v[j] + (1.f - b1) * gf;
        vv.v[j] = b2 * vv.v[j] + (1.f - b2) * gf * gf;
        float upd = (mv.v[j] / bc1) / (sqrtf(vv.v[j] / bc2) + eps);
        pv.v[j] -= lr * upd;
      }
      *reinterpret_cast<float4v*>(p + i) = pv;
      *reinterpret_cast<float4v*>(m + i) = mv;
      *reinterpret_cast<float4v*>(v + i) = vv;
      if (out16) {
#pragma unroll
        for (int j = 0; j < 4; ++j) out16[i + j] = (P16)pv.v[j];
      }
    } else {
      for (int64_t k = i; k < n; ++k) {
        float gf = (float)g[k] + wd * p[k];
        m[k] = b1 * m[k] + (1.f - b1) * gf;
        v[k] = b2 * v[k] + (1.f - b2) * gf * gf;
        p[k] -= lr * (m[k] / bc1) / (sqrtf(v[k] / bc2) + eps);
        if (out16) out16[k] = (P16)p[k];
      }
    }
  }
}
}  // namespace

void adam_step(torch::Tensor param32, torch::Tensor grad, torch::Tensor m,
               torch::Tensor v, double lr, double beta1, double beta2,
               double eps, double weight_decay, int64_t step,
               torch::Tensor out16, torch::Tensor bc_dev) {
  const int64_t n = param32.numel();
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>((n + BLOCK * 4 - 1) / (BLOCK * 4), 4096);
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2 = 1.f - powf((float)beta2, (float)step);
  bool has16 = out16.numel() > 0;
  TORCH_CHECK(param32.scalar_type() == at::kFloat, "master must be fp32");
  DISPATCH_FLOAT(grad, "adam_step", [&] {
    if (has16) {
      TORCH_CHECK(out16.scalar_type() == at::kBFloat16,
                  "adam out16 must be bf16");
      hipLaunchKernelGGL((adam_kernel<scalar_t, bf16>), dim3(grid),
                         dim3(BLOCK), 0, stream,
                         param32.data_ptr<float>(),
                         (const scalar_t*)grad.data_ptr(),
                         m.data_ptr<float>(), v.data_ptr<float>(),
                         (bf16*)out16.data_ptr(), n, (float)lr,
                         (float)beta1, (float)beta2, (float)eps,
                         (float)weight_decay, bc1, bc2,
                         bc_dev.numel() ? bc_dev.data_ptr<float>()
                                        : nullptr);
    } else {
      hipLaunchKernelGGL((adam_kernel<scalar_t, float>), dim3(grid),
                         dim3(BLOCK), 0, stream,
                         param32.data_ptr<float>(),
                         (const scalar_t*)grad.data_ptr(),
                         m.data_ptr<float>(), v.data_ptr<float>(),
                         (float*)nullptr, n, (float)lr, (float)beta1,
                         (float)beta2, (float)eps, (float)weight_decay,
                         bc1, bc2,
                         bc_dev.numel() ? bc_dev.data_ptr<float>()
                                        : nullptr);
    }
  });
}
