// Python bindings for the mi355 CDNA4 kernel library (built by
// tools/build_ext.py with hipcc --offload-arch=gfx950).
#include <torch/extension.h>

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                                       torch::Tensor inv, c10::optional<torch::Tensor> dextra,
                                       double p_drop, c10::optional<torch::Tensor> seed,
                                       long salt);
std::vector<torch::Tensor> add_rmsnorm_fwd(torch::Tensor x, torch::Tensor r,
                                           torch::Tensor w, double eps,
                                           double p_drop, c10::optional<torch::Tensor> seed,
                                           long salt);
torch::Tensor geglu_fwd(torch::Tensor gate, torch::Tensor up);
torch::Tensor geglu_packed_fwd(torch::Tensor x, double p_drop,
                               c10::optional<torch::Tensor> seed, long salt);
torch::Tensor geglu_packed_bwd(torch::Tensor dout, torch::Tensor x, double p_drop,
                               c10::optional<torch::Tensor> seed, long salt);
std::vector<torch::Tensor> geglu_bwd(torch::Tensor dout, torch::Tensor gate, torch::Tensor up);
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels, long ignore_index);
torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor labels, torch::Tensor lse,
                     long ignore_index, torch::Tensor dloss, torch::Tensor nvalid);
void sumsq_into(torch::Tensor x, torch::Tensor out);
void adamw_multi(std::vector<torch::Tensor> masters, std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs,
                 std::vector<torch::Tensor> lowps, double lr, double beta1, double beta2,
                 double eps, double weight_decay, long step, torch::Tensor grad_scale);
torch::Tensor decode_attn(torch::Tensor q, torch::Tensor kc, torch::Tensor vc,
                          long cache_len, c10::optional<torch::Tensor> bias,
                          c10::optional<torch::Tensor> kpad);
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    c10::optional<torch::Tensor> bias,
                                    c10::optional<torch::Tensor> kpad, bool causal,
                                    double p_drop, c10::optional<torch::Tensor> seed);
torch::Tensor token_embed_fwd(torch::Tensor ids, torch::Tensor weight);
torch::Tensor token_embed_bwd(torch::Tensor ids, torch::Tensor dy, long vocab,
                              torch::ScalarType out_dtype);
torch::Tensor rel_bias_fwd(torch::Tensor buckets, torch::Tensor weight);
torch::Tensor rel_bias_bwd(torch::Tensor buckets, torch::Tensor dbias, long NB);
std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o, torch::Tensor lse,
                                    c10::optional<torch::Tensor> bias,
                                    c10::optional<torch::Tensor> kpad,
                                    bool causal, bool need_dbias,
                                    double p_drop, c10::optional<torch::Tensor> seed,
                                    c10::optional<torch::Tensor> packed);
torch::Tensor seg_postprocess(torch::Tensor logits, long Ho, long Wo);
torch::Tensor upsample_bilinear_fwd(torch::Tensor x, long Ho, long Wo);
torch::Tensor upsample_bilinear_bwd(torch::Tensor dout, long Hi, long Wi);
torch::Tensor gemm_nt(torch::Tensor a, torch::Tensor b, long splitk);
torch::Tensor dwconv3x3_fwd(torch::Tensor x, torch::Tensor w,
                            c10::optional<torch::Tensor> bias, bool flip);
std::vector<torch::Tensor> dwconv3x3_wrw(torch::Tensor dout, torch::Tensor x);
torch::Tensor patch_embed_s1_fwd(torch::Tensor x, torch::Tensor w,
                                 c10::optional<torch::Tensor> bias);
std::vector<torch::Tensor> patch_embed_s1_wrw(torch::Tensor dout, torch::Tensor x);
std::vector<torch::Tensor> seg_ce_fwd(torch::Tensor logits, torch::Tensor labels,
                                      long Ho, long Wo, long ignore_index);
torch::Tensor seg_ce_bwd(torch::Tensor logits, torch::Tensor labels,
                         torch::Tensor lse, torch::Tensor gout,
                         torch::Tensor nvalid, long ignore_index);
torch::Tensor gemm_nn(torch::Tensor a, torch::Tensor b, long splitk);
torch::Tensor gemm_tn(torch::Tensor a, torch::Tensor b, long splitk);
std::vector<torch::Tensor> attn_fwd_v2(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                       c10::optional<torch::Tensor> bias,
                                       c10::optional<torch::Tensor> kpad, bool causal,
                                       double p_drop, c10::optional<torch::Tensor> seed);
std::vector<torch::Tensor> attn_bwd_v2(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                       torch::Tensor v, torch::Tensor o, torch::Tensor lse,
                                       c10::optional<torch::Tensor> bias,
                                       c10::optional<torch::Tensor> kpad,
                                       bool causal, bool need_dbias,
                                       double p_drop, c10::optional<torch::Tensor> seed);
std::vector<torch::Tensor> attn_bwd_v3(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                       torch::Tensor v, torch::Tensor o, torch::Tensor lse,
                                       c10::optional<torch::Tensor> bias,
                                       c10::optional<torch::Tensor> kpad,
                                       bool causal, bool need_dbias,
                                       double p_drop, c10::optional<torch::Tensor> seed,
                                       c10::optional<torch::Tensor> packed);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("add_rmsnorm_fwd", &add_rmsnorm_fwd);
  m.def("geglu_fwd", &geglu_fwd);
  m.def("geglu_packed_fwd", &geglu_packed_fwd);
  m.def("geglu_packed_bwd", &geglu_packed_bwd);
  m.def("geglu_bwd", &geglu_bwd);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("adamw_multi", &adamw_multi);
  m.def("sumsq_into", &sumsq_into);
  m.def("decode_attn", &decode_attn);
  m.def("attn_fwd", &attn_fwd);
  m.def("token_embed_fwd", &token_embed_fwd);
  m.def("token_embed_bwd", &token_embed_bwd);
  m.def("rel_bias_fwd", &rel_bias_fwd);
  m.def("rel_bias_bwd", &rel_bias_bwd);
  m.def("attn_bwd", &attn_bwd, py::arg("dout"), py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("o"), py::arg("lse"), py::arg("bias"),
        py::arg("kpad"), py::arg("causal"), py::arg("need_dbias"),
        py::arg("p_drop"), py::arg("seed"), py::arg("packed") = c10::nullopt);
  m.def("seg_postprocess", &seg_postprocess);
  m.def("upsample_bilinear_fwd", &upsample_bilinear_fwd);
  m.def("upsample_bilinear_bwd", &upsample_bilinear_bwd);
  m.def("gemm_nt", &gemm_nt, py::arg("a"), py::arg("b"), py::arg("splitk") = 0);
  m.def("dwconv3x3_fwd", &dwconv3x3_fwd, py::arg("x"), py::arg("w"),
        py::arg("bias") = c10::nullopt, py::arg("flip") = false);
  m.def("dwconv3x3_wrw", &dwconv3x3_wrw);
  m.def("patch_embed_s1_fwd", &patch_embed_s1_fwd, py::arg("x"), py::arg("w"),
        py::arg("bias") = c10::nullopt);
  m.def("patch_embed_s1_wrw", &patch_embed_s1_wrw);
  m.def("seg_ce_fwd", &seg_ce_fwd);
  m.def("seg_ce_bwd", &seg_ce_bwd);
  m.def("gemm_nn", &gemm_nn, py::arg("a"), py::arg("b"), py::arg("splitk") = 0);
  m.def("gemm_tn", &gemm_tn, py::arg("a"), py::arg("b"), py::arg("splitk") = 0);
  m.def("attn_fwd_v2", &attn_fwd_v2);
  m.def("attn_bwd_v2", &attn_bwd_v2);
  m.def("attn_bwd_v3", &attn_bwd_v3, py::arg("dout"), py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("o"), py::arg("lse"), py::arg("bias"),
        py::arg("kpad"), py::arg("causal"), py::arg("need_dbias"),
        py::arg("p_drop"), py::arg("seed"), py::arg("packed") = c10::nullopt);
  m.attr("__gfx__") = "gfx950";
}
