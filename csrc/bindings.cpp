// Python bindings for the gfx950 kernel library (_hip_ops).
#include <torch/extension.h>

#include <vector>

// elementwise.hip
std::vector<at::Tensor> layernorm_fwd(const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&, double);
std::vector<at::Tensor> add_layernorm_fwd(const at::Tensor&, const at::Tensor&,
                                          const at::Tensor&, const at::Tensor&,
                                          double);
std::vector<at::Tensor> layernorm_bwd(const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&);
at::Tensor bias_gelu_fwd(const at::Tensor&, const at::Tensor&);
std::vector<at::Tensor> bias_gelu_bwd(const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&);
std::vector<at::Tensor> masked_ce_fwd(const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&);
at::Tensor masked_ce_bwd(const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, const at::Tensor&,
                         const at::Tensor&);
at::Tensor colsum(const at::Tensor&);
std::vector<at::Tensor> dropout_add_ln_fwd(const at::Tensor&,
                                           const at::Tensor&,
                                           const at::Tensor&,
                                           const at::Tensor&, double, double,
                                           const at::Tensor&);
at::Tensor mask_scale(const at::Tensor&, const at::Tensor&, double);
void bump_counter(const at::Tensor&);
at::Tensor wgrad(const at::Tensor&, const at::Tensor&, long);
at::Tensor wgrad2(const at::Tensor&, const at::Tensor&, long);
at::Tensor gemm_nt(const at::Tensor&, const at::Tensor&,
                   c10::optional<at::Tensor>, bool);
at::Tensor embed3_fwd(const at::Tensor&, const at::Tensor&,
                      const at::Tensor&, const at::Tensor&,
                      const at::Tensor&);
// crf.hip
std::vector<at::Tensor> crf_fwd(const at::Tensor&, const at::Tensor&,
                                const at::Tensor&, const at::Tensor&);
at::Tensor crf_viterbi(const at::Tensor&, const at::Tensor&,
                       const at::Tensor&);
// softlexicon.hip
at::Tensor softlexicon_fwd(const at::Tensor&, const at::Tensor&,
                           const at::Tensor&);
std::vector<at::Tensor> softlexicon_bwd(const at::Tensor&, const at::Tensor&,
                                        const at::Tensor&, const at::Tensor&);
// adam.hip
void multi_tensor_adamw(std::vector<at::Tensor>, std::vector<at::Tensor>,
                        std::vector<at::Tensor>, std::vector<at::Tensor>,
                        std::vector<at::Tensor>, std::vector<double>,
                        std::vector<double>, double, double, double,
                        c10::optional<at::Tensor>);
at::Tensor multi_tensor_sumsq(std::vector<at::Tensor>);
void multi_tensor_scale(std::vector<at::Tensor>, const at::Tensor&);
std::vector<at::Tensor> adamw_build_meta(
    std::vector<at::Tensor>, std::vector<at::Tensor>, std::vector<at::Tensor>,
    std::vector<at::Tensor>, std::vector<at::Tensor>, std::vector<double>,
    std::vector<double>);
void multi_tensor_adamw_run(const at::Tensor&, long, long, bool, double,
                            double, double, c10::optional<at::Tensor>);
std::vector<at::Tensor> norm_build_meta(std::vector<at::Tensor>);
at::Tensor multi_tensor_sumsq_run(const at::Tensor&, long, const at::Tensor&);
void multi_tensor_scale_run(const at::Tensor&, long, const at::Tensor&);
// attention.hip
std::vector<at::Tensor> attn_fwd(const at::Tensor&, const at::Tensor&,
                                 const at::Tensor&, const at::Tensor&, double,
                                 double, c10::optional<at::Tensor>);
std::vector<at::Tensor> attn_bwd(const at::Tensor&, const at::Tensor&,
                                 const at::Tensor&, const at::Tensor&,
                                 const at::Tensor&, const at::Tensor&,
                                 const at::Tensor&, double, double,
                                 c10::optional<at::Tensor>);
std::vector<at::Tensor> attn_fwd_qkv(const at::Tensor&, const at::Tensor&,
                                     double, double,
                                     c10::optional<at::Tensor>);
std::vector<at::Tensor> attn_bwd_qkv(const at::Tensor&, const at::Tensor&,
                                     const at::Tensor&, const at::Tensor&,
                                     const at::Tensor&, double, double,
                                     c10::optional<at::Tensor>);
// tener.hip
std::vector<at::Tensor> tener_attn_fwd(const at::Tensor&, const at::Tensor&,
                                       const at::Tensor&, const at::Tensor&,
                                       const at::Tensor&, const at::Tensor&);
std::vector<at::Tensor> tener_attn_bwd(const at::Tensor&, const at::Tensor&,
                                       const at::Tensor&, const at::Tensor&,
                                       const at::Tensor&, const at::Tensor&,
                                       const at::Tensor&, const at::Tensor&,
                                       const at::Tensor&);
// probe.hip
at::Tensor mfma_probe(const at::Tensor&, const at::Tensor&);
// lstm.hip
std::vector<at::Tensor> lstm_fwd(const at::Tensor&, const at::Tensor&,
                                 const at::Tensor&, bool, bool);
std::vector<at::Tensor> lstm_bwd(const at::Tensor&, const at::Tensor&,
                                 const at::Tensor&, const at::Tensor&,
                                 const at::Tensor&, const at::Tensor&, bool,
                                 bool);
std::vector<at::Tensor> bilstm_fwd_l(const at::Tensor&, const at::Tensor&,
                                     const at::Tensor&, bool, double);
at::Tensor bilstm_bwd_l(const at::Tensor&, const at::Tensor&,
                        const at::Tensor&, const at::Tensor&,
                        const at::Tensor&, bool, double);

PYBIND11_MODULE(_hip_ops, m) {
  m.doc() = "chinesener_amd gfx950 HIP kernels";
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("add_layernorm_fwd", &add_layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("bias_gelu_fwd", &bias_gelu_fwd);
  m.def("bias_gelu_bwd", &bias_gelu_bwd);
  m.def("masked_ce_fwd", &masked_ce_fwd);
  m.def("masked_ce_bwd", &masked_ce_bwd);
  m.def("colsum", &colsum);
  m.def("dropout_add_ln_fwd", &dropout_add_ln_fwd);
  m.def("mask_scale", &mask_scale);
  m.def("bump_counter", &bump_counter);
  m.def("wgrad", &wgrad);
  m.def("wgrad2", &wgrad2);
  m.def("gemm_nt", &gemm_nt, py::arg("a"), py::arg("b"),
        py::arg("bias") = c10::nullopt, py::arg("fp32_out") = false);
  m.def("embed3_fwd", &embed3_fwd);
  m.def("crf_fwd", &crf_fwd);
  m.def("crf_viterbi", &crf_viterbi);
  m.def("softlexicon_fwd", &softlexicon_fwd);
  m.def("softlexicon_bwd", &softlexicon_bwd);
  m.def("multi_tensor_adamw", &multi_tensor_adamw);
  m.def("multi_tensor_sumsq", &multi_tensor_sumsq);
  m.def("multi_tensor_scale", &multi_tensor_scale);
  m.def("adamw_build_meta", &adamw_build_meta);
  m.def("multi_tensor_adamw_run", &multi_tensor_adamw_run);
  m.def("norm_build_meta", &norm_build_meta);
  m.def("multi_tensor_sumsq_run", &multi_tensor_sumsq_run);
  m.def("multi_tensor_scale_run", &multi_tensor_scale_run);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("attn_fwd_qkv", &attn_fwd_qkv);
  m.def("attn_bwd_qkv", &attn_bwd_qkv);
  m.def("tener_attn_fwd", &tener_attn_fwd);
  m.def("tener_attn_bwd", &tener_attn_bwd);
  m.def("mfma_probe", &mfma_probe);
  m.def("lstm_fwd", &lstm_fwd);
  m.def("lstm_bwd", &lstm_bwd);
  m.def("bilstm_fwd", &bilstm_fwd_l);
  m.def("bilstm_bwd", &bilstm_bwd_l);
}
