// Embedding lookup fwd (row gather) + bwd (scatter-add into fp32, cast).
// Reference parity: hetu/impl/kernel/EmbeddingLookup.cu.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {
constexpr int BLOCK = 256;

template <typename T>
__global__ void embedding_fwd_kernel(const T* __restrict__ table,
                                     const int64_t* __restrict__ ids,
                                     T* __restrict__ y, int64_t n, int D) {
  constexpr int V = VecIO<T>::VEC;
  for (int64_t row = blockIdx.x; row < n; row += gridDim.x) {
    const T* src = table + ids[row] * D;
    T* dst = y + row * D;
    for (int i = threadIdx.x * V; i + V <= D; i += BLOCK * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(src + i, v);
      VecIO<T>::store(dst + i, v);
    }
    for (int i = (D / V) * V + threadIdx.x; i < D; i += BLOCK)
      dst[i] = src[i];
  }
}

template <typename T>
__global__ void embedding_bwd_kernel(const T* __restrict__ dy,
                                     const int64_t* __restrict__ ids,
                                     float* __restrict__ acc,
                                     int64_t n, int D) {
  for (int64_t row = blockIdx.x; row < n; row += gridDim.x) {
    const T* src = dy + row * D;
    float* dst = acc + ids[row] * D;
    for (int i = threadIdx.x; i < D; i += BLOCK)
      atomicAdd(dst + i, (float)src[i]);
  }
}

__global__ void cast_kernel_bf16(const float* __restrict__ in,
                                 bf16* __restrict__ out, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * BLOCK;
  for (; i < n; i += stride) out[i] = (bf16)in[i];
}
}  // namespace

torch::Tensor embedding_fwd(torch::Tensor table, torch::Tensor ids) {
  const int D = table.size(1);
  const int64_t n = ids.numel();
  auto sizes = ids.sizes().vec();
  sizes.push_back(D);
  auto y = torch::empty(sizes, table.options());
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(n, 16384);
  DISPATCH_FLOAT(table, "embedding_fwd", [&] {
    hipLaunchKernelGGL(embedding_fwd_kernel<scalar_t>, dim3(grid),
                       dim3(BLOCK), 0, stream,
                       (const scalar_t*)table.data_ptr(),
                       ids.data_ptr<int64_t>(), (scalar_t*)y.data_ptr(),
                       n, D);
  });
  return y;
}

torch::Tensor embedding_bwd(torch::Tensor dy, torch::Tensor ids,
                            int64_t num_rows) {
  const int D = dy.size(-1);
  const int64_t n = ids.numel();
  auto acc = torch::zeros({num_rows, D}, dy.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(n, 16384);
  DISPATCH_FLOAT(dy, "embedding_bwd", [&] {
    hipLaunchKernelGGL(embedding_bwd_kernel<scalar_t>, dim3(grid),
                       dim3(BLOCK), 0, stream,
                       (const scalar_t*)dy.data_ptr(),
                       ids.data_ptr<int64_t>(), acc.data_ptr<float>(),
                       n, D);
  });
  return acc.to(dy.scalar_type());
}
