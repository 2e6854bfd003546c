// SwiGLU fused fwd/bwd: y = silu(x1) * x2 over last-dim halves.
// Reference parity: hetu/impl/kernel/SwiGLU.cu:14,30. HBM-bound: bf16x8
// vector IO, grid-stride over rows.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {
constexpr int BLOCK = 256;

template <typename T>
__global__ void swiglu_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                  int64_t rows, int F) {
  constexpr int V = VecIO<T>::VEC;
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* x1 = x + row * 2 * F;
    const T* x2 = x1 + F;
    T* yr = y + row * F;
    for (int i = threadIdx.x * V; i < F; i += BLOCK * V) {
      float a[VecIO<T>::VEC], b[VecIO<T>::VEC];
      VecIO<T>::load(x1 + i, a);
      VecIO<T>::load(x2 + i, b);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float s = 1.f / (1.f + __expf(-a[j]));
        a[j] = a[j] * s * b[j];
      }
      VecIO<T>::store(yr + i, a);
    }
  }
}

template <typename T>
__global__ void swiglu_bwd_kernel(const T* __restrict__ dy,
                                  const T* __restrict__ x,
                                  T* __restrict__ dx, int64_t rows, int F) {
  constexpr int V = VecIO<T>::VEC;
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* x1 = x + row * 2 * F;
    const T* x2 = x1 + F;
    const T* dyr = dy + row * F;
    T* dx1 = dx + row * 2 * F;
    T* dx2 = dx1 + F;
    for (int i = threadIdx.x * V; i < F; i += BLOCK * V) {
      float a[VecIO<T>::VEC], b[VecIO<T>::VEC], g[VecIO<T>::VEC];
      float o1[VecIO<T>::VEC], o2[VecIO<T>::VEC];
      VecIO<T>::load(x1 + i, a);
      VecIO<T>::load(x2 + i, b);
      VecIO<T>::load(dyr + i, g);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float s = 1.f / (1.f + __expf(-a[j]));
        float silu = a[j] * s;
        float dsilu = s * (1.f + a[j] * (1.f - s));
        o1[j] = g[j] * b[j] * dsilu;
        o2[j] = g[j] * silu;
      }
      VecIO<T>::store(dx1 + i, o1);
      VecIO<T>::store(dx2 + i, o2);
    }
  }
}
}  // namespace

torch::Tensor swiglu_fwd(torch::Tensor x) {
  const int F = x.size(-1) / 2;
  const int64_t rows = x.numel() / (2 * F);
  TORCH_CHECK(F % 8 == 0, "swiglu: half-dim must be a multiple of 8");
  auto sizes = x.sizes().vec();
  sizes.back() = F;
  auto y = torch::empty(sizes, x.options());
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(rows, 8192);
  DISPATCH_FLOAT(x, "swiglu_fwd", [&] {
    hipLaunchKernelGGL(swiglu_fwd_kernel<scalar_t>, dim3(grid), dim3(BLOCK),
                       0, stream, (const scalar_t*)x.data_ptr(),
                       (scalar_t*)y.data_ptr(), rows, F);
  });
  return y;
}

torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor x) {
  const int F = x.size(-1) / 2;
  const int64_t rows = x.numel() / (2 * F);
  auto dx = torch::empty_like(x);
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(rows, 8192);
  DISPATCH_FLOAT(x, "swiglu_bwd", [&] {
    hipLaunchKernelGGL(swiglu_bwd_kernel<scalar_t>, dim3(grid), dim3(BLOCK),
                       0, stream, (const scalar_t*)dy.data_ptr(),
                       (const scalar_t*)x.data_ptr(),
                       (scalar_t*)dx.data_ptr(), rows, F);
  });
  return dx;
}
