// Blockwise quantization kernels: int8 / fp4 / nf4 (+ dequant).
// Reference parity: hetu/impl/kernel/quantization.cu:13-89 and
// graph/ops/Quantization.h (bitsandbytes-style blockwise absmax
// quantization backing the 4-bit matmul path).  CDNA4-native: one wave
// handles whole blocks; bf16/f32 IO vectorized; the 16-entry fp4/nf4
// codebooks live in registers (no LDS needed).
#include <torch/extension.h>
#include "ext_stream.h"
#include "common.h"

namespace {

// fp4 (e2m1) and nf4 codebooks (bitsandbytes layouts)
__constant__ float FP4_CODE[16] = {
    0.0f, 0.0052083333f, 0.6666667f, 1.0f, 0.3333333f, 0.5f,
    0.1666667f, 0.25f,
    -0.0f, -0.0052083333f, -0.6666667f, -1.0f, -0.3333333f, -0.5f,
    -0.1666667f, -0.25f};
__constant__ float NF4_CODE[16] = {
    -1.0f, -0.6961928009986877f, -0.5250730514526367f,
    -0.39491748809814453f, -0.28444138169288635f, -0.18477343022823334f,
    -0.09105003625154495f, 0.0f, 0.07958029955625534f,
    0.16093020141124725f, 0.24611230194568634f, 0.33791524171829224f,
    0.44070982933044434f, 0.5626170039176941f, 0.7229568362236023f, 1.0f};

template <typename T>
__global__ void quant8_kernel(const T* __restrict__ x,
                              unsigned char* __restrict__ q,
                              float* __restrict__ absmax,
                              int64_t n, int bs) {
  const int64_t blk = blockIdx.x;
  const int64_t base = blk * bs;
  if (base >= n) return;
  const int m = (int)min((int64_t)bs, n - base);
  __shared__ float sm[16];
  float amax = 0.f;
  for (int i = threadIdx.x; i < m; i += blockDim.x)
    amax = fmaxf(amax, fabsf((float)x[base + i]));
  amax = block_max(amax, sm);
  if (threadIdx.x == 0) absmax[blk] = amax;
  const float inv = amax > 0 ? 127.f / amax : 0.f;
  for (int i = threadIdx.x; i < m; i += blockDim.x) {
    int v = (int)lrintf((float)x[base + i] * inv);
    q[base + i] = (unsigned char)(v + 128);
  }
}

template <typename T>
__global__ void dequant8_kernel(const unsigned char* __restrict__ q,
                                const float* __restrict__ absmax,
                                T* __restrict__ y, int64_t n, int bs) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  float s = absmax[i / bs] / 127.f;
  y[i] = (T)(((int)q[i] - 128) * s);
}

template <typename T, bool NF4>
__global__ void quant4_kernel(const T* __restrict__ x,
                              unsigned char* __restrict__ q,
                              float* __restrict__ absmax,
                              int64_t n, int bs) {
  const int64_t blk = blockIdx.x;
  const int64_t base = blk * bs;
  if (base >= n) return;
  const int m = (int)min((int64_t)bs, n - base);
  __shared__ float sm[16];
  float amax = 0.f;
  for (int i = threadIdx.x; i < m; i += blockDim.x)
    amax = fmaxf(amax, fabsf((float)x[base + i]));
  amax = block_max(amax, sm);
  if (threadIdx.x == 0) absmax[blk] = amax;
  const float inv = amax > 0 ? 1.f / amax : 0.f;
  const float* code = NF4 ? NF4_CODE : FP4_CODE;
  // two values per byte; each thread packs element pairs
  for (int i = threadIdx.x * 2; i < m; i += blockDim.x * 2) {
    unsigned char byte = 0;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      float v = (i + j < m) ? (float)x[base + i + j] * inv : 0.f;
      int best = 0;
      float bd = 1e30f;
#pragma unroll
      for (int c = 0; c < 16; ++c) {
        float d = fabsf(v - code[c]);
        if (d < bd) { bd = d; best = c; }
      }
      byte |= (unsigned char)best << (j == 0 ? 4 : 0);
    }
    q[(base + i) / 2] = byte;
  }
}

template <typename T, bool NF4>
__global__ void dequant4_kernel(const unsigned char* __restrict__ q,
                                const float* __restrict__ absmax,
                                T* __restrict__ y, int64_t n, int bs) {
  int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 2;
  if (i >= n) return;
  const float* code = NF4 ? NF4_CODE : FP4_CODE;
  float s = absmax[i / bs];
  unsigned char b = q[i / 2];
  y[i] = (T)(code[b >> 4] * s);
  if (i + 1 < n) {
    float s2 = absmax[(i + 1) / bs];
    y[i + 1] = (T)(code[b & 15] * s2);
  }
}

}  // namespace

std::vector<torch::Tensor> quantize_blockwise(torch::Tensor x,
                                              std::string qtype,
                                              int64_t blocksize) {
  TORCH_CHECK(x.is_contiguous());
  int64_t n = x.numel();
  int64_t nblk = (n + blocksize - 1) / blocksize;
  auto absmax = torch::empty({nblk}, x.options().dtype(at::kFloat));
  auto stream = hetu_current_stream();
  if (qtype == "int8") {
    auto q = torch::empty({n}, x.options().dtype(at::kByte));
    DISPATCH_FLOAT(x, "quant8", [&] {
      hipLaunchKernelGGL(quant8_kernel<scalar_t>, dim3(nblk), dim3(256), 0,
                         stream, (const scalar_t*)x.data_ptr(),
                         q.data_ptr<unsigned char>(),
                         absmax.data_ptr<float>(), n, (int)blocksize);
    });
    return {q, absmax};
  }
  TORCH_CHECK(qtype == "fp4" || qtype == "nf4", "qtype int8|fp4|nf4");
  TORCH_CHECK(blocksize % 2 == 0 && n % 2 == 0);
  auto q = torch::empty({n / 2}, x.options().dtype(at::kByte));
  bool nf4 = qtype == "nf4";
  DISPATCH_FLOAT(x, "quant4", [&] {
    if (nf4) {
      hipLaunchKernelGGL((quant4_kernel<scalar_t, true>), dim3(nblk),
                         dim3(256), 0, stream,
                         (const scalar_t*)x.data_ptr(),
                         q.data_ptr<unsigned char>(),
                         absmax.data_ptr<float>(), n, (int)blocksize);
    } else {
      hipLaunchKernelGGL((quant4_kernel<scalar_t, false>), dim3(nblk),
                         dim3(256), 0, stream,
                         (const scalar_t*)x.data_ptr(),
                         q.data_ptr<unsigned char>(),
                         absmax.data_ptr<float>(), n, (int)blocksize);
    }
  });
  return {q, absmax};
}

torch::Tensor dequantize_blockwise(torch::Tensor q, torch::Tensor absmax,
                                   std::string qtype, int64_t blocksize,
                                   int64_t numel,
                                   torch::ScalarType out_dtype) {
  auto y = torch::empty({numel}, q.options().dtype(out_dtype));
  auto stream = hetu_current_stream();
  if (qtype == "int8") {
    int64_t grid = (numel + 255) / 256;
    DISPATCH_FLOAT(y, "dequant8", [&] {
      hipLaunchKernelGGL(dequant8_kernel<scalar_t>, dim3(grid), dim3(256),
                         0, stream, q.data_ptr<unsigned char>(),
                         absmax.data_ptr<float>(),
                         (scalar_t*)y.data_ptr(), numel, (int)blocksize);
    });
    return y;
  }
  bool nf4 = qtype == "nf4";
  int64_t grid = (numel / 2 + 255) / 256;
  DISPATCH_FLOAT(y, "dequant4", [&] {
    if (nf4) {
      hipLaunchKernelGGL((dequant4_kernel<scalar_t, true>), dim3(grid),
                         dim3(256), 0, stream,
                         q.data_ptr<unsigned char>(),
                         absmax.data_ptr<float>(), (scalar_t*)y.data_ptr(),
                         numel, (int)blocksize);
    } else {
      hipLaunchKernelGGL((dequant4_kernel<scalar_t, false>), dim3(grid),
                         dim3(256), 0, stream,
                         q.data_ptr<unsigned char>(),
                         absmax.data_ptr<float>(), (scalar_t*)y.data_ptr(),
                         numel, (int)blocksize);
    }
  });
  return y;
}
