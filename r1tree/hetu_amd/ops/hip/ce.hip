// Sparse softmax cross-entropy + vocab-parallel CE local stage.
// Reference parity: SoftmaxCrossEntropySparse.cu,
// VocabParallelCrossEntropyLoss.cu:15,70 (block per row over vocab shard;
// cross-rank max/sum allreduce happens at op level).
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {
constexpr int BLOCK = 256;

template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              float* __restrict__ loss,
                              float* __restrict__ lse_out,
                              int64_t rows, int V_, int64_t ignore) {
  constexpr int V = VecIO<T>::VEC;
  __shared__ float smem[16];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * V_;
    float mx = -INFINITY;
    for (int i = threadIdx.x * V; i + V <= V_; i += BLOCK * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(lr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) mx = fmaxf(mx, v[j]);
    }
    for (int i = (V_ / V) * V + threadIdx.x; i < V_; i += BLOCK)
      mx = fmaxf(mx, (float)lr[i]);
    mx = block_max(mx, smem);
    float sum = 0.f;
    for (int i = threadIdx.x * V; i + V <= V_; i += BLOCK * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(lr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) sum += __expf(v[j] - mx);
    }
    for (int i = (V_ / V) * V + threadIdx.x; i < V_; i += BLOCK)
      sum += __expf((float)lr[i] - mx);
    sum = block_sum(sum, smem);
    if (threadIdx.x == 0) {
      float lse = mx + __logf(sum);
      lse_out[row] = lse;
      int64_t lbl = labels[row];
      loss[row] = (lbl == ignore) ? 0.f : lse - (float)lr[lbl];
    }
  }
}

template <typename T>
__global__ void ce_bwd_kernel(const float* __restrict__ dloss,
                              const T* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              const float* __restrict__ lse,
                              T* __restrict__ dlogits,
                              int64_t rows, int V_, int64_t ignore) {
  constexpr int V = VecIO<T>::VEC;
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * V_;
    T* dr = dlogits + row * V_;
    const int64_t lbl = labels[row];
    const float g = (lbl == ignore) ? 0.f : dloss[row];
    const float l = lse[row];
    for (int i = threadIdx.x * V; i + V <= V_; i += BLOCK * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(lr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float p = __expf(v[j] - l);
        v[j] = g * (p - ((i + j) == lbl ? 1.f : 0.f));
      }
      VecIO<T>::store(dr + i, v);
    }
    for (int i = (V_ / V) * V + threadIdx.x; i < V_; i += BLOCK) {
      float p = __expf((float)lr[i] - l);
      dr[i] = (T)(g * (p - (i == lbl ? 1.f : 0.f)));
    }
  }
}

// vocab-parallel local stage: per-row local max and (masked) predicted logit
template <typename T>
__global__ void vp_ce_local_kernel(const T* __restrict__ logits,
                                   const int64_t* __restrict__ labels,
                                   float* __restrict__ lmax,
                                   float* __restrict__ picked,
                                   int64_t rows, int Vloc,
                                   int64_t vstart, int64_t vend,
                                   int64_t ignore) {
  constexpr int V = VecIO<T>::VEC;
  __shared__ float smem[16];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * Vloc;
    float mx = -INFINITY;
    for (int i = threadIdx.x * V; i + V <= Vloc; i += BLOCK * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(lr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) mx = fmaxf(mx, v[j]);
    }
    for (int i = (Vloc / V) * V + threadIdx.x; i < Vloc; i += BLOCK)
      mx = fmaxf(mx, (float)lr[i]);
    mx = block_max(mx, smem);
    if (threadIdx.x == 0) {
      lmax[row] = mx;
      int64_t lbl = labels[row];
      bool own = lbl >= vstart && lbl < vend && lbl != ignore;
      picked[row] = own ? (float)lr[lbl - vstart] : 0.f;
    }
  }
}
}  // namespace

std::vector<torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                          torch::Tensor labels,
                                          int64_t ignore_index) {
  const int V_ = logits.size(-1);
  const int64_t rows = logits.numel() / V_;
  auto loss = torch::empty({rows}, logits.options().dtype(at::kFloat));
  auto lse = torch::empty({rows}, logits.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(rows, 8192);
  DISPATCH_FLOAT(logits, "softmax_ce_fwd", [&] {
    hipLaunchKernelGGL(ce_fwd_kernel<scalar_t>, dim3(grid), dim3(BLOCK), 0,
                       stream, (const scalar_t*)logits.data_ptr(),
                       labels.data_ptr<int64_t>(), loss.data_ptr<float>(),
                       lse.data_ptr<float>(), rows, V_, ignore_index);
  });
  auto row_sizes = logits.sizes().vec();
  row_sizes.pop_back();
  return {loss.view(row_sizes), lse.view(row_sizes)};
}

torch::Tensor softmax_ce_bwd(torch::Tensor dloss, torch::Tensor logits,
                             torch::Tensor labels, torch::Tensor lse,
                             int64_t ignore_index) {
  const int V_ = logits.size(-1);
  const int64_t rows = logits.numel() / V_;
  auto dlogits = torch::empty_like(logits);
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(rows, 8192);
  auto dl = dloss.to(at::kFloat).contiguous();
  DISPATCH_FLOAT(logits, "softmax_ce_bwd", [&] {
    hipLaunchKernelGGL(ce_bwd_kernel<scalar_t>, dim3(grid), dim3(BLOCK), 0,
                       stream, dl.data_ptr<float>(),
                       (const scalar_t*)logits.data_ptr(),
                       labels.data_ptr<int64_t>(), lse.data_ptr<float>(),
                       (scalar_t*)dlogits.data_ptr(), rows, V_,
                       ignore_index);
  });
  return dlogits;
}

std::vector<torch::Tensor> vp_ce_local(torch::Tensor logits,
                                       torch::Tensor labels,
                                       int64_t vocab_start,
                                       int64_t vocab_end,
                                       int64_t ignore_index) {
  const int Vloc = logits.size(-1);
  const int64_t rows = logits.numel() / Vloc;
  auto lmax = torch::empty({rows}, logits.options().dtype(at::kFloat));
  auto picked = torch::empty({rows}, logits.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(rows, 8192);
  DISPATCH_FLOAT(logits, "vp_ce_local", [&] {
    hipLaunchKernelGGL(vp_ce_local_kernel<scalar_t>, dim3(grid), dim3(BLOCK),
                       0, stream, (const scalar_t*)logits.data_ptr(),
                       labels.data_ptr<int64_t>(), lmax.data_ptr<float>(),
                       picked.data_ptr<float>(), rows, Vloc, vocab_start,
                       vocab_end, ignore_index);
  });
  auto row_sizes = logits.sizes().vec();
  row_sizes.pop_back();
  return {lmax.view(row_sizes), picked.view(row_sizes)};
}

namespace {

// sum of exp(logit - gmax[row]) straight off bf16 (no fp32 logits copy)
template <typename T>
__global__ void vp_sumexp_kernel(const T* __restrict__ logits,
                                 const float* __restrict__ gmax,
                                 float* __restrict__ gsum,
                                 int64_t rows, int Vloc) {
  constexpr int V = VecIO<T>::VEC;
  __shared__ float smem[16];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * Vloc;
    const float mx = gmax[row];
    float s = 0.f;
    for (int i = threadIdx.x * V; i + V <= Vloc; i += BLOCK * V) {
      float v[VecIO<T>::VEC];
      VecIO<T>::load(lr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) s += __expf(v[j] - mx);
    }
    for (int i = (Vloc / V) * V + threadIdx.x; i < Vloc; i += BLOCK)
      s += __expf((float)lr[i] - mx);
    s = block_sum(s, smem);
    if (threadIdx.x == 0) gsum[row] = s;
  }
}

// dlogits = gy * (softmax - onehot_local) with the GLOBAL lse; rows whose
// label is out of this shard still get the softmax term
template <typename T>
__global__ void vp_ce_bwd_kernel(const float* __restrict__ gy,
                                 const T* __restrict__ logits,
                                 const int64_t* __restrict__ labels,
                                 const float* __restrict__ lse,
                                 T* __restrict__ dl, int64_t rows, int Vloc,
                                 int64_t vstart, int64_t vend,
                                 int64_t ignore) {
  constexpr int V = VecIO<T>::VEC;
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* lr = logits + row * Vloc;
    T* dr = dl + row * Vloc;
    const int64_t lbl = labels[row];
    const float g = (lbl == ignore) ? 0.f : gy[row];
    const float l = lse[row];
    const int64_t local = (lbl >= vstart && lbl < vend) ? lbl - vstart : -1;
    for (int i = threadIdx.x * V; i + V <= Vloc; i += BLOCK * V) {
      float v[VecIO<T>::VEC], o[VecIO<T>::VEC];
      VecIO<T>::load(lr + i, v);
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float p = __expf(v[j] - l);
        o[j] = g * (p - ((i + j == local) ? 1.f : 0.f));
      }
      VecIO<T>::store(dr + i, o);
    }
    for (int i = (Vloc / V) * V + threadIdx.x; i < Vloc; i += BLOCK) {
      float p = __expf((float)lr[i] - l);
      dr[i] = (T)(g * (p - ((i == local) ? 1.f : 0.f)));
    }
  }
}

}  // namespace

torch::Tensor vp_sumexp(torch::Tensor logits, torch::Tensor gmax) {
  const int Vloc = logits.size(-1);
  const int64_t rows = logits.numel() / Vloc;
  auto gsum = torch::empty({rows}, logits.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(rows, 8192);
  auto gm = gmax.contiguous();
  DISPATCH_FLOAT(logits, "vp_sumexp", [&] {
    hipLaunchKernelGGL(vp_sumexp_kernel<scalar_t>, dim3(grid), dim3(BLOCK),
                       0, stream, (const scalar_t*)logits.data_ptr(),
                       gm.data_ptr<float>(), gsum.data_ptr<float>(), rows,
                       Vloc);
  });
  return gsum;
}

torch::Tensor vp_ce_bwd(torch::Tensor gy, torch::Tensor logits,
                        torch::Tensor labels, torch::Tensor lse,
                        int64_t vstart, int64_t vend, int64_t ignore) {
  const int Vloc = logits.size(-1);
  const int64_t rows = logits.numel() / Vloc;
  auto dl = torch::empty_like(logits);
  auto stream = hetu_current_stream();
  int grid = (int)std::min<int64_t>(rows, 8192);
  auto gyf = gy.to(at::kFloat).contiguous();
  auto lc = lse.to(at::kFloat).contiguous();
  DISPATCH_FLOAT(logits, "vp_ce_bwd", [&] {
    hipLaunchKernelGGL(vp_ce_bwd_kernel<scalar_t>, dim3(grid), dim3(BLOCK),
                       0, stream, gyf.data_ptr<float>(),
                       (const scalar_t*)logits.data_ptr(),
                       labels.data_ptr<int64_t>(), lc.data_ptr<float>(),
                       (scalar_t*)dl.data_ptr(), rows, Vloc, vstart, vend,
                       ignore);
  });
  return dl;
}
