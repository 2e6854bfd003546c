// RoPE fwd/bwd (NeoX half-rotation) for x [B, S, H, D], cos/sin [S, D/2].
// Reference parity: hetu/impl/kernel/rotary.cu:97-185. One thread per
// (token-head, d/2 pair chunk); cos/sin read once per row from L2.
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {
constexpr int BLOCK = 256;

// sign=+1 forward, -1 backward (rotation by -angle)
template <typename T, int SIGN>
__global__ void rope_kernel(const T* __restrict__ x,
                            const float* __restrict__ cs,
                            const float* __restrict__ sn,
                            T* __restrict__ y,
                            int64_t nrows,   // B*S*H
                            int S, int H, int D,
                            // token stride in elements: rows of the same
                            // token are adjacent (head-major within a
                            // token); contiguous [B,S,H,D] has rs = H*D
                            int64_t rs) {
  constexpr int V = 4;  // process 4 (d, d+D/2) pairs per iteration
  const int half = D / 2;
  for (int64_t row = blockIdx.x; row < nrows; row += gridDim.x) {
    const int s = (int)((row / H) % S);
    const int64_t off = (row / H) * rs + (row % H) * D;
    const T* xr = x + off;
    T* yr = y + off;
    const float* c = cs + (int64_t)s * half;
    const float* sn_r = sn + (int64_t)s * half;
    for (int i = threadIdx.x * V; i < half; i += BLOCK * V) {
      float x1[V], x2[V], cc[V], ss[V];
#pragma unroll
      for (int j = 0; j < V; ++j) {
        x1[j] = (float)xr[i + j];
        x2[j] = (float)xr[i + j + half];
        cc[j] = c[i + j];
        ss[j] = SIGN * sn_r[i + j];
      }
#pragma unroll
      for (int j = 0; j < V; ++j) {
        yr[i + j] = (T)(x1[j] * cc[j] - x2[j] * ss[j]);
        yr[i + j + half] = (T)(x2[j] * cc[j] + x1[j] * ss[j]);
      }
    }
  }
}

template <int SIGN>
torch::Tensor rope_run(torch::Tensor x, torch::Tensor cs, torch::Tensor sn) {
  TORCH_CHECK(x.dim() >= 3, "rope expects [..., S, H, D]");
  const int D = x.size(-1);
  const int H = x.size(-2);
  const int S = x.size(-3);
  const int64_t nrows = x.numel() / D;
  TORCH_CHECK(cs.scalar_type() == at::kFloat, "rope cos must be fp32");
  auto y = torch::empty_like(x);
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(nrows, 16384);
  DISPATCH_FLOAT(x, "rope", [&] {
    hipLaunchKernelGGL((rope_kernel<scalar_t, SIGN>), dim3(grid),
                       dim3(BLOCK), 0, stream,
                       (const scalar_t*)x.data_ptr(),
                       cs.data_ptr<float>(), sn.data_ptr<float>(),
                       (scalar_t*)y.data_ptr(), nrows, S, H, D,
                       (int64_t)H * D);
  });
  return y;
}
}  // namespace

// In-place rotation of the first `n_rot` heads of every token row of a
// fused [B, S, C] qkv buffer (q|k sections are adjacent, so q+k rotate in
// ONE launch; rotation is linear so backward never needs the pre-rotation
// values).  sign=+1 fwd, -1 bwd.
void rope_qk_inplace(torch::Tensor qkv, torch::Tensor cs, torch::Tensor sn,
                     int64_t n_rot, int64_t D, int64_t sign) {
  TORCH_CHECK(qkv.dim() == 3 && qkv.is_contiguous());
  TORCH_CHECK(cs.scalar_type() == at::kFloat);
  const int S = qkv.size(1);
  const int64_t C = qkv.size(2);
  const int64_t nrows = (int64_t)qkv.size(0) * S * n_rot;
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(nrows, 16384);
  DISPATCH_FLOAT(qkv, "rope_qk", [&] {
    if (sign > 0) {
      hipLaunchKernelGGL((rope_kernel<scalar_t, 1>), dim3(grid),
                         dim3(BLOCK), 0, stream,
                         (const scalar_t*)qkv.data_ptr(),
                         cs.data_ptr<float>(), sn.data_ptr<float>(),
                         (scalar_t*)qkv.data_ptr(), nrows, S, (int)n_rot,
                         (int)D, C);
    } else {
      hipLaunchKernelGGL((rope_kernel<scalar_t, -1>), dim3(grid),
                         dim3(BLOCK), 0, stream,
                         (const scalar_t*)qkv.data_ptr(),
                         cs.data_ptr<float>(), sn.data_ptr<float>(),
                         (scalar_t*)qkv.data_ptr(), nrows, S, (int)n_rot,
                         (int)D, C);
    }
  });
}

torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cs, torch::Tensor sn) {
  return rope_run<1>(x, cs, sn);
}

torch::Tensor rope_bwd(torch::Tensor dy, torch::Tensor cs, torch::Tensor sn) {
  return rope_run<-1>(dy, cs, sn);
}
