// Python bindings for the nanorlhf_amd gfx950 HIP kernel library.
#include <torch/extension.h>

// elementwise.hip
std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor invrms);
std::vector<torch::Tensor> rmsnorm_addres_fwd(torch::Tensor x, torch::Tensor res,
                                              torch::Tensor w, double eps);
torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor table, torch::Tensor positions, double sign);
torch::Tensor swiglu_fwd(torch::Tensor gu);
torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor gu);
// logprob.hip
void ce_rowstats(torch::Tensor logits, torch::Tensor labels, double inv_temp,
                 torch::Tensor lp, torch::Tensor ent, torch::Tensor lse);
void ce_backward_dlogits(torch::Tensor logits, torch::Tensor labels,
                         torch::Tensor lse, torch::Tensor g, double inv_temp);
// adamw.hip
void adamw_step(torch::Tensor p, torch::Tensor g, torch::Tensor m, torch::Tensor v,
                double lr, double b1, double b2, double eps, double wd, long step);
// sampling.hip
torch::Tensor sample_topp(torch::Tensor logits, double temperature, double top_p,
                          long seed, long step);
std::vector<torch::Tensor> sample_topp_dev(torch::Tensor logits, double temperature,
                                           double top_p, long seed, torch::Tensor step);
// kvcache.hip
void kv_append(torch::Tensor k, torch::Tensor v, torch::Tensor slots,
               torch::Tensor k_cache, torch::Tensor v_cache);
torch::Tensor paged_attn_decode(torch::Tensor q, torch::Tensor k_cache,
                                torch::Tensor v_cache, torch::Tensor block_tables,
                                torch::Tensor seq_lens, double scale);
// masked.hip
std::vector<torch::Tensor> masked_moments(torch::Tensor v, torch::Tensor mask);
torch::Tensor whiten_apply(torch::Tensor v, double mean, double invstd, double shift);
// lora.hip
torch::Tensor lora_gemm(torch::Tensor x, torch::Tensor w,
                        c10::optional<torch::Tensor> u,
                        c10::optional<torch::Tensor> b,
                        c10::optional<torch::Tensor> bias);
void lora_add_(torch::Tensor y, torch::Tensor u, torch::Tensor b);
// attention.hip
std::vector<torch::Tensor> fa_fwd_varlen(torch::Tensor q, torch::Tensor k,
                                         torch::Tensor v, torch::Tensor cu_seqlens,
                                         long max_seqlen, double scale, bool causal);
std::vector<torch::Tensor> fa_bwd_varlen(torch::Tensor dout, torch::Tensor q,
                                         torch::Tensor k, torch::Tensor v,
                                         torch::Tensor o, torch::Tensor lse,
                                         torch::Tensor cu_seqlens, long max_seqlen,
                                         double scale, bool causal);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("rmsnorm_addres_fwd", &rmsnorm_addres_fwd);
  m.def("rope_fwd", &rope_fwd);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("ce_rowstats", &ce_rowstats);
  m.def("ce_backward_dlogits", &ce_backward_dlogits);
  m.def("adamw_step", &adamw_step);
  m.def("sample_topp", &sample_topp);
  m.def("sample_topp_dev", &sample_topp_dev);
  m.def("kv_append", &kv_append);
  m.def("paged_attn_decode", &paged_attn_decode);
  m.def("masked_moments", &masked_moments);
  m.def("whiten_apply", &whiten_apply);
  m.def("lora_gemm", &lora_gemm, py::arg("x"), py::arg("w"),
        py::arg("u") = py::none(), py::arg("b") = py::none(),
        py::arg("bias") = py::none());
  m.def("lora_add_", &lora_add_);
  m.def("fa_fwd_varlen", &fa_fwd_varlen);
  m.def("fa_bwd_varlen", &fa_bwd_varlen);
}
