// Stateless Philox dropout fwd/bwd (reference Dropout.cu): the mask is
// reproducible from (seed, offset, index), so backward can either reuse the
// saved mask (returned as uint8) or regenerate; we save the mask (cheap,
// 1 B/elem) to keep backward a pure elementwise pass.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {
constexpr int BLOCK = 256;

template <typename T>
__global__ void dropout_fwd_kernel(const T* __restrict__ x,
                                   T* __restrict__ y,
                                   unsigned char* __restrict__ mask,
                                   int64_t n, float p, float inv_keep,
                                   u64 seed, u64 offset) {
  int64_t idx = ((int64_t)blockIdx.x * BLOCK + threadIdx.x) * 4;
  const int64_t stride = (int64_t)gridDim.x * BLOCK * 4;
  for (; idx < n; idx += stride) {
    Philox ph(seed, offset, (u32)(idx >> 2));
    u32 r[4];
    ph.next4(r);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      if (idx + j < n) {
        bool keep = u32_to_uniform(r[j]) > p;
        mask[idx + j] = keep;
        y[idx + j] = keep ? (T)((float)x[idx + j] * inv_keep) : (T)0.f;
      }
    }
  }
}

template <typename T>
__global__ void dropout_bwd_kernel(const T* __restrict__ dy,
                                   const unsigned char* __restrict__ mask,
                                   T* __restrict__ dx, int64_t n,
                                   float inv_keep) {
  int64_t idx = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (; idx < n; idx += stride)
    dx[idx] = mask[idx] ? (T)((float)dy[idx] * inv_keep) : (T)0.f;
}
}  // namespace

std::vector<torch::Tensor> dropout_fwd(torch::Tensor x, double p,
                                       int64_t seed, int64_t offset) {
  const int64_t n = x.numel();
  auto y = torch::empty_like(x);
  auto mask = torch::empty_like(x, x.options().dtype(at::kByte));
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>((n + BLOCK * 4 - 1) / (BLOCK * 4), 4096);
  DISPATCH_FLOAT(x, "dropout_fwd", [&] {
    hipLaunchKernelGGL(dropout_fwd_kernel<scalar_t>, dim3(grid), dim3(BLOCK),
                       0, stream, (const scalar_t*)x.data_ptr(),
                       (scalar_t*)y.data_ptr(),
                       mask.data_ptr<unsigned char>(), n, (float)p,
                       (float)(1.0 / (1.0 - p)), (u64)seed, (u64)offset);
  });
  return {y, mask};
}

torch::Tensor dropout_bwd(torch::Tensor dy, torch::Tensor mask, double p,
                          int64_t seed, int64_t offset) {
  const int64_t n = dy.numel();
  auto dx = torch::empty_like(dy);
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>((n + BLOCK - 1) / BLOCK, 8192);
  DISPATCH_FLOAT(dy, "dropout_bwd", [&] {
    hipLaunchKernelGGL(dropout_bwd_kernel<scalar_t>, dim3(grid), dim3(BLOCK),
                       0, stream, (const scalar_t*)dy.data_ptr(),
                       mask.data_ptr<unsigned char>(),
                       (scalar_t*)dx.data_ptr(), n,
                       (float)(1.0 / (1.0 - p)));
  });
  return dx;
}
