// Fused activation fwd/bwd kernels: gelu (tanh approx), silu.
// Reference parity: hetu/impl/kernel/Gelu.cu, Activation.cu — rewritten for
// CDNA4: HBM-bound elementwise, bf16x8 vector IO (16 B/lane), grid-stride.
// The backward is a single fused kernel (the torch fallback was a ~7-kernel
// fp32 chain costing ~13% of a GPT step — see profiles/r01_*).
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {
constexpr int BLOCK = 256;

DEV float gelu_f(float x) {
  const float c = 0.7978845608028654f;  // sqrt(2/pi)
  const float a = 0.044715f;
  float t = tanhf(c * (x + a * x * x * x));
  return 0.5f * x * (1.f + t);
}

DEV float gelu_df(float x) {
  const float c = 0.7978845608028654f;
  const float a = 0.044715f;
  float t = tanhf(c * (x + a * x * x * x));
  float dt = (1.f - t * t) * c * (1.f + 3.f * a * x * x);
  return 0.5f * (1.f + t) + 0.5f * x * dt;
}

DEV float silu_f(float x) {
  float s = 1.f / (1.f + __expf(-x));
  return x * s;
}

DEV float silu_df(float x) {
  float s = 1.f / (1.f + __expf(-x));
  return s * (1.f + x * (1.f - s));
}

#define UNARY_KERNELS(NAME, FWD, DF)                                         \
  template <typename T>                                                      \
  __global__ void NAME##_fwd_kernel(const T* __restrict__ x,                 \
                                    T* __restrict__ y, int64_t n) {          \
    constexpr int V = VecIO<T>::VEC;                                         \
    int64_t i0 = ((int64_t)blockIdx.x * BLOCK + threadIdx.x) * V;            \
    int64_t stride = (int64_t)gridDim.x * BLOCK * V;                         \
    for (int64_t i = i0; i < n; i += stride) {                               \
      float a[VecIO<T>::VEC];                                                \
      VecIO<T>::load(x + i, a);                                              \
      _Pragma("unroll") for (int j = 0; j < V; ++j) a[j] = FWD(a[j]);        \
      VecIO<T>::store(y + i, a);                                             \
    }                                                                        \
  }                                                                          \
  template <typename T>                                                      \
  __global__ void NAME##_bwd_kernel(const T* __restrict__ dy,                \
                                    const T* __restrict__ x,                 \
                                    T* __restrict__ dx, int64_t n) {         \
    constexpr int V = VecIO<T>::VEC;                                         \
    int64_t i0 = ((int64_t)blockIdx.x * BLOCK + threadIdx.x) * V;            \
    int64_t stride = (int64_t)gridDim.x * BLOCK * V;                         \
    for (int64_t i = i0; i < n; i += stride) {                               \
      float a[VecIO<T>::VEC], g[VecIO<T>::VEC];                              \
      VecIO<T>::load(x + i, a);                                              \
      VecIO<T>::load(dy + i, g);                                             \
      _Pragma("unroll") for (int j = 0; j < V; ++j) a[j] = g[j] * DF(a[j]);  \
      VecIO<T>::store(dx + i, a);                                            \
    }                                                                        \
  }

UNARY_KERNELS(gelu, gelu_f, gelu_df)
UNARY_KERNELS(silu, silu_f, silu_df)

int pick_grid(int64_t n, int vec) {
  int64_t blocks = (n / vec + BLOCK - 1) / BLOCK;
  // >=2048 workgroups fills 256 CUs across the 8 XCDs
  return (int)std::min<int64_t>(blocks, 16384);
}
}  // namespace

#define UNARY_API(NAME)                                                      \
  torch::Tensor NAME##_fwd(torch::Tensor x) {                                \
    TORCH_CHECK(x.is_contiguous());                                          \
    int64_t n = x.numel();                                                   \
    auto y = torch::empty_like(x);                                           \
    auto stream = hetu_current_stream();                                     \
    DISPATCH_FLOAT(x, #NAME "_fwd", [&] {                                    \
      TORCH_CHECK(n % VecIO<scalar_t>::VEC == 0,                             \
                  #NAME ": numel must be a multiple of the vector width");   \
      hipLaunchKernelGGL(NAME##_fwd_kernel<scalar_t>,                        \
                         dim3(pick_grid(n, VecIO<scalar_t>::VEC)),           \
                         dim3(BLOCK), 0, stream,                             \
                         (const scalar_t*)x.data_ptr(),                      \
                         (scalar_t*)y.data_ptr(), n);                        \
    });                                                                      \
    return y;                                                                \
  }                                                                          \
  torch::Tensor NAME##_bwd(torch::Tensor dy, torch::Tensor x) {              \
    TORCH_CHECK(x.is_contiguous() && dy.is_contiguous());                    \
    int64_t n = x.numel();                                                   \
    auto dx = torch::empty_like(x);                                          \
    auto stream = hetu_current_stream();                                     \
    DISPATCH_FLOAT(x, #NAME "_bwd", [&] {                                    \
      TORCH_CHECK(n % VecIO<scalar_t>::VEC == 0,                             \
                  #NAME ": numel must be a multiple of the vector width");   \
      hipLaunchKernelGGL(NAME##_bwd_kernel<scalar_t>,                        \
                         dim3(pick_grid(n, VecIO<scalar_t>::VEC)),           \
                         dim3(BLOCK), 0, stream,                             \
                         (const scalar_t*)dy.data_ptr(),                     \
                         (const scalar_t*)x.data_ptr(),                      \
                         (scalar_t*)dx.data_ptr(), n);                       \
    });                                                                      \
    return dx;                                                               \
  }

UNARY_API(gelu)
UNARY_API(silu)
