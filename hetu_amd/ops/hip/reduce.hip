// Column-sum reduction kernels (bias gradients): out[c] = sum_r in[r, c].
//
// Replaces at::native's two-pass reduce on the hot path.  Besides being a
// hand CDNA4 kernel (coalesced vectorized rows, fp32 accumulation), it is
// hipGraph-replay-safe by construction: stage 1 writes disjoint per-row-
// block partials, stage 2 combines them in a fixed order — deterministic,
// no semaphore/workspace state carried between runs.  (Observed on ROCm
// 7.2: the at::native column-sum inside a captured train step returns
// garbage from the 2nd replay on; see profiles/r02_capture_replay_bug.md.)
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include "common.h"
#include "ext_stream.h"

namespace {

constexpr int CBLOCK = 256;

template <typename T>
__global__ void colsum_part_kernel(const T* __restrict__ in,
                                   float* __restrict__ part,
                                   int64_t R, int64_t C,
                                   int64_t rows_per_blk) {
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
    float* dst = part + (int64_t)blockIdx.y * C + c0;
#pragma unroll
    for (int j = 0; j < V; ++j) dst[j] = acc[j];
  } else {
    for (int64_t c = c0; c < C; ++c) {
      float a = 0.f;
      for (int64_t r = r0; r < r1; ++r)
        a += (float)in[r * C + c];
      part[(int64_t)blockIdx.y * C + c] = a;
    }
  }
}

__global__ void colsum_combine_kernel(const float* __restrict__ part,
                                      float* __restrict__ out,
                                      int64_t C, int ny) {
  int64_t c = (int64_t)blockIdx.x * CBLOCK + threadIdx.x;
  if (c >= C) return;
  float a = 0.f;
  for (int y = 0; y < ny; ++y)
    a += part[(int64_t)y * C + c];
  out[c] = a;
}

}  // namespace

torch::Tensor colsum(torch::Tensor x) {
  TORCH_CHECK(x.dim() == 2 && x.is_contiguous(), "colsum: 2-D contiguous");
  const int64_t R = x.size(0), C = x.size(1);
  auto stream = hetu_current_stream();
  auto out = torch::empty({C}, x.options().dtype(at::kFloat));
  DISPATCH_FLOAT(x, "colsum", [&] {
    constexpr int V = VecIO<scalar_t>::VEC;
    int64_t gx = (C + (int64_t)CBLOCK * V - 1) / ((int64_t)CBLOCK * V);
    // fill the chip: aim for >=1024 blocks total via row splits
    int64_t gy = std::min<int64_t>(
        std::max<int64_t>(1024 / std::max<int64_t>(gx, 1), 1),
        std::max<int64_t>(R / 16, 1));
    int64_t rows_per_blk = (R + gy - 1) / gy;
    gy = (R + rows_per_blk - 1) / rows_per_blk;
    if (gy == 1) {
      hipLaunchKernelGGL((colsum_part_kernel<scalar_t>), dim3(gx, 1),
                         dim3(CBLOCK), 0, stream,
                         (const scalar_t*)x.data_ptr(),
                         out.data_ptr<float>(), R, C, R);
    } else {
      auto part = torch::empty({gy, C}, x.options().dtype(at::kFloat));
      hipLaunchKernelGGL((colsum_part_kernel<scalar_t>), dim3(gx, gy),
                         dim3(CBLOCK), 0, stream,
                         (const scalar_t*)x.data_ptr(),
                         part.data_ptr<float>(), R, C, rows_per_blk);
      int64_t gc = (C + CBLOCK - 1) / CBLOCK;
      hipLaunchKernelGGL(colsum_combine_kernel, dim3(gc), dim3(CBLOCK), 0,
                         stream, part.data_ptr<float>(),
                         out.data_ptr<float>(), C, (int)gy);
    }
  });
  return out;
}
