// pybind bindings for the _hetu_hip extension (gfx950 kernels).
#include <torch/extension.h>

#include <vector>

// norms.hip
std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd);
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd);
// reduce.hip
torch::Tensor colsum(torch::Tensor x);
// ltgemm.cpp
std::vector<torch::Tensor> lt_linear_gelu_aux(torch::Tensor x,
                                              torch::Tensor w,
                                              torch::Tensor b);
std::vector<torch::Tensor> lt_dgelu_bgrad(torch::Tensor dy, torch::Tensor w,
                                          torch::Tensor aux);
// swiglu.hip
torch::Tensor swiglu_fwd(torch::Tensor x);
torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor x);
// activations.hip
torch::Tensor gelu_fwd(torch::Tensor x);
torch::Tensor gelu_bwd(torch::Tensor dy, torch::Tensor x);
torch::Tensor silu_fwd(torch::Tensor x);
torch::Tensor silu_bwd(torch::Tensor dy, torch::Tensor x);
// rope.hip
torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cos, torch::Tensor sin);
torch::Tensor rope_bwd(torch::Tensor dy, torch::Tensor cos, torch::Tensor sin);
// softmax.hip
torch::Tensor softmax_fwd(torch::Tensor x);
torch::Tensor softmax_bwd(torch::Tensor dy, torch::Tensor y);
// ce.hip
std::vector<torch::Tensor> softmax_ce_fwd(torch::Tensor logits,
                                          torch::Tensor labels,
                                          int64_t ignore_index);
torch::Tensor softmax_ce_bwd(torch::Tensor dloss, torch::Tensor logits,
                             torch::Tensor labels, torch::Tensor lse,
                             int64_t ignore_index);
torch::Tensor vp_sumexp(torch::Tensor logits, torch::Tensor gmax);
torch::Tensor vp_ce_bwd(torch::Tensor gy, torch::Tensor logits,
                        torch::Tensor labels, torch::Tensor lse,
                        int64_t vstart, int64_t vend, int64_t ignore);
std::vector<torch::Tensor> vp_ce_local(torch::Tensor logits,
                                       torch::Tensor labels,
                                       int64_t vocab_start,
                                       int64_t vocab_end,
                                       int64_t ignore_index);
// dropout.hip
std::vector<torch::Tensor> dropout_fwd(torch::Tensor x, double p,
                                       int64_t seed, int64_t offset);
torch::Tensor dropout_bwd(torch::Tensor dy, torch::Tensor mask, double p,
                          int64_t seed, int64_t offset);
// embedding.hip
torch::Tensor embedding_fwd(torch::Tensor table, torch::Tensor ids);
torch::Tensor embedding_bwd(torch::Tensor dy, torch::Tensor ids,
                            int64_t num_rows);
std::vector<torch::Tensor> rmsnorm_fwd_res(torch::Tensor x,
                                           torch::Tensor resid,
                                           torch::Tensor w, double eps);
std::vector<torch::Tensor> rmsnorm_bwd2_res(torch::Tensor dy,
                                            torch::Tensor s,
                                            torch::Tensor w,
                                            torch::Tensor rstd,
                                            torch::Tensor ds_ext);
std::vector<torch::Tensor> layernorm_fwd_res(torch::Tensor x,
                                             torch::Tensor resid,
                                             torch::Tensor w,
                                             torch::Tensor b, double eps);
std::vector<torch::Tensor> layernorm_bwd2_res(torch::Tensor dy,
                                              torch::Tensor s,
                                              torch::Tensor w,
                                              torch::Tensor mean,
                                              torch::Tensor rstd,
                                              torch::Tensor ds_ext);
std::vector<torch::Tensor> layernorm_bwd2(torch::Tensor dy, torch::Tensor x,
                                          torch::Tensor w,
                                          torch::Tensor mean,
                                          torch::Tensor rstd);
std::vector<torch::Tensor> rmsnorm_bwd2(torch::Tensor dy, torch::Tensor x,
                                        torch::Tensor w,
                                        torch::Tensor rstd);
// adam.hip
void adam_step(torch::Tensor param32, torch::Tensor grad, torch::Tensor m,
               torch::Tensor v, double lr, double beta1, double beta2,
               double eps, double weight_decay, int64_t step,
               torch::Tensor out16, torch::Tensor bc_dev);
// gemm.hip
torch::Tensor gemm_bf16(torch::Tensor x, torch::Tensor w, bool trans_w);
// quant.hip
std::vector<torch::Tensor> quantize_blockwise(torch::Tensor x,
                                              std::string qtype,
                                              int64_t blocksize);
torch::Tensor dequantize_blockwise(torch::Tensor q, torch::Tensor absmax,
                                   std::string qtype, int64_t blocksize,
                                   int64_t numel,
                                   torch::ScalarType out_dtype);
// galvatron_dp.cpp
std::pair<double, std::vector<int64_t>> galvatron_dp(
    std::vector<double> times, std::vector<double> mems, int64_t L,
    int64_t S, double cap, int64_t buckets);
// attention.hip
std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, bool causal,
                                          double scale);
std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor dout,
                                          torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, torch::Tensor out,
                                          torch::Tensor lse, bool causal,
                                          double scale);
// fused-qkv fast path (attention_v2 / attention_bwd_v2)
std::vector<torch::Tensor> flash_attn_fwd_qkv(torch::Tensor qkv, int64_t H,
                                              int64_t Hkv, int64_t D,
                                              bool causal, double scale);
torch::Tensor flash_attn_bwd_qkv(torch::Tensor dout, torch::Tensor qkv,
                                 torch::Tensor out, torch::Tensor lse,
                                 int64_t H, int64_t Hkv, int64_t D,
                                 bool causal, double scale);
// rope.hip
void rope_qk_inplace(torch::Tensor qkv, torch::Tensor cs, torch::Tensor sn,
                     int64_t n_rot, int64_t D, int64_t sign);
// attention_v3.hip (experimental round-2 candidate)
std::vector<torch::Tensor> flash_attn_varlen_fwd(
    torch::Tensor q, torch::Tensor k, torch::Tensor v, torch::Tensor cu,
    int64_t max_seqlen, bool causal, double scale);
std::vector<torch::Tensor> flash_attn_fwd_v3(torch::Tensor q,
                                             torch::Tensor k,
                                             torch::Tensor v, bool causal,
                                             double scale);
std::vector<torch::Tensor> flash_attn_bwd_v3(torch::Tensor dout,
                                             torch::Tensor q,
                                             torch::Tensor k,
                                             torch::Tensor v,
                                             torch::Tensor out,
                                             torch::Tensor lse,
                                             bool causal, double scale);

// embed_cache.cpp
void register_embed_cache(pybind11::module& m);
// dataloader.cpp
void register_dataloader(pybind11::module& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  register_embed_cache(m);
  register_dataloader(m);
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("layernorm_bwd2", &layernorm_bwd2);
  m.def("layernorm_fwd_res", &layernorm_fwd_res);
  m.def("rmsnorm_fwd_res", &rmsnorm_fwd_res);
  m.def("rmsnorm_bwd2_res", &rmsnorm_bwd2_res);
  m.def("layernorm_bwd2_res", &layernorm_bwd2_res);
  m.def("rmsnorm_bwd2", &rmsnorm_bwd2);
  m.def("colsum", &colsum);
  m.def("lt_linear_gelu_aux", &lt_linear_gelu_aux);
  m.def("lt_dgelu_bgrad", &lt_dgelu_bgrad);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("gelu_fwd", &gelu_fwd);
  m.def("gelu_bwd", &gelu_bwd);
  m.def("silu_fwd", &silu_fwd);
  m.def("silu_bwd", &silu_bwd);
  m.def("rope_fwd", &rope_fwd);
  m.def("rope_bwd", &rope_bwd);
  m.def("softmax_fwd", &softmax_fwd);
  m.def("softmax_bwd", &softmax_bwd);
  m.def("softmax_ce_fwd", &softmax_ce_fwd);
  m.def("softmax_ce_bwd", &softmax_ce_bwd);
  m.def("vp_ce_local", &vp_ce_local);
  m.def("vp_sumexp", &vp_sumexp);
  m.def("vp_ce_bwd", &vp_ce_bwd);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("embedding_fwd", &embedding_fwd);
  m.def("embedding_bwd", &embedding_bwd);
  m.def("adam_step", &adam_step);
  m.def("gemm_bf16", &gemm_bf16);
  m.def("quantize_blockwise", &quantize_blockwise);
  m.def("dequantize_blockwise", &dequantize_blockwise);
  m.def("galvatron_dp", &galvatron_dp);
  m.def("flash_attn_fwd", &flash_attn_fwd);
  m.def("flash_attn_bwd", &flash_attn_bwd);
  m.def("flash_attn_fwd_qkv", &flash_attn_fwd_qkv);
  m.def("flash_attn_varlen_fwd", &flash_attn_varlen_fwd);
  m.def("flash_attn_bwd_qkv", &flash_attn_bwd_qkv);
  m.def("rope_qk_inplace", &rope_qk_inplace);
  m.def("flash_attn_fwd_v3", &flash_attn_fwd_v3);
  m.def("flash_attn_bwd_v3", &flash_attn_bwd_v3);
}
