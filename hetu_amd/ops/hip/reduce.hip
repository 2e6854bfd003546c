// Column-sum reduction kernels (bias gradients): out[c] = sum_r in[r, c].
//
// Replaces at::native's two-pass reduce on the hot path.  Besides being a
// hand CDNA4 kernel (coalesced vectorized rows, fp32 accumulation), it is
// hipGraph-replay-safe by construction: the accumulator is zeroed by a
// CAPTURED memset (torch::zeros) and cross-block combination uses plain
// fp32 atomicAdd — no semaphore/workspace state carried between runs.
// (Observed on ROCm 7.2: the at::native column-sum for e.g. [512, 1024]
// bf16 inside a captured train step returns garbage from the 2nd replay
// on; see profiles/r02_capture_replay_bug.md.)
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include "common.h"
#include "ext_stream.h"

namespace {

constexpr int CBLOCK = 256;

template <typename T>
__global__ void colsum_kernel(const T* __restrict__ in,
                              float* __restrict__ out,
                              int64_t R, int64_t C, int64_t rows_per_blk) {
  constexpr int V = VecIO<T>::VEC;
  int64_t c0 = ((int64_t)blockIdx.x * CBLOCK + threadIdx.x) * V;
  if (c0 >= C) return;
  int64_t r0 = (int64_t)blockIdx.y * rows_per_blk;
  int64_t r1 = min(r0 + rows_per_blk, R);
  float acc[VecIO<T>::VEC];
#pragma unroll
  for (int j = 0; j < V; ++j) acc[j] = 0.f;
  if (c0 + V <= C) {
    for (int64_t r = r0; r < r1; ++r) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(in + r * C + c0, v);
#pragma unroll
      for (int j = 0; j < V; ++j) acc[j] += v[j];
    }
    if (gridDim.y == 1) {
#pragma unroll
      for (int j = 0; j < V; ++j) out[c0 + j] = acc[j];
    } else {
#pragma unroll
      for (int j = 0; j < V; ++j) atomicAdd(out + c0 + j, acc[j]);
    }
  } else {
    // ragged tail: scalar columns
    for (int64_t c = c0; c < C; ++c) {
      float a = 0.f;
      for (int64_t r = r0; r < r1; ++r)
        a += (float)in[r * C + c];
      if (gridDim.y == 1) out[c] = a;
      else atomicAdd(out + c, a);
    }
  }
}

}  // namespace

torch::Tensor colsum(torch::Tensor x) {
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous(), "colsum: 2-D contiguous");
  const int64_t R = x.size(0), C = x.size(1);
  auto stream = hetu_current_stream();
  int64_t gx = 0, gy = 0, rows_per_blk = 0;
  torch::Tensor out;
  DISPATCH_FLOAT(x, "colsum", [&] {
    constexpr int V = VecIO<scalar_t>::VEC;
    gx = (C + (int64_t)CBLOCK * V - 1) / ((int64_t)CBLOCK * V);
    // fill the chip: aim for >=1024 blocks total via row splits
    gy = std::min<int64_t>(std::max<int64_t>(1024 / std::max<int64_t>(gx, 1),
                                             1),
                           std::max<int64_t>(R / 16, 1));
    rows_per_blk = (R + gy - 1) / gy;
    // zeroed only when atomics accumulate across row-splits
    out = (gy > 1) ? torch::zeros({C}, x.options().dtype(at::kFloat))
                   : torch::empty({C}, x.options().dtype(at::kFloat));
    hipLaunchKernelGGL((colsum_kernel<scalar_t>), dim3(gx, gy),
                       dim3(CBLOCK), 0, stream,
                       (const scalar_t*)x.data_ptr(), out.data_ptr<float>(),
                       R, C, rows_per_blk);
  });
  return out;
}
