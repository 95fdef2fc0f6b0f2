// Python bindings for the jimm_amd HIP/CDNA4 extension.

#include <torch/extension.h>

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b, double eps);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w,
                                         torch::Tensor mean, torch::Tensor rstd,
                                         c10::optional<torch::Tensor> addend);
torch::Tensor bias_act_fwd(torch::Tensor z, c10::optional<torch::Tensor> bias, std::string act,
                           c10::optional<torch::Tensor> residual);
std::vector<torch::Tensor> bias_act_fwd_fp8(torch::Tensor z, c10::optional<torch::Tensor> bias,
                                            std::string act, torch::Tensor scale8,
                                            torch::Tensor amax);
std::vector<torch::Tensor> gemm_nt_8p_gradact_fp8(torch::Tensor dy, torch::Tensor wt,
                                                  torch::Tensor z, std::string act,
                                                  torch::Tensor scale8, torch::Tensor amax);
std::vector<torch::Tensor> layernorm_fwd_fp8(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                                             double eps, torch::Tensor scale8,
                                             torch::Tensor amax);
torch::Tensor act_bwd(torch::Tensor dy, torch::Tensor z, std::string act);
torch::Tensor colsum(torch::Tensor dz);
torch::Tensor im2col_patch(torch::Tensor img, int64_t patch);
torch::Tensor cls_pos_fwd(torch::Tensor x, c10::optional<torch::Tensor> cls, torch::Tensor pos);
torch::Tensor embed_pos_fwd(torch::Tensor ids, torch::Tensor emb, torch::Tensor pos);
torch::Tensor col2im_patch(torch::Tensor cols, std::vector<int64_t> img_shape, int64_t patch);
void adam_step(std::vector<torch::Tensor> ps, std::vector<torch::Tensor> gs,
               std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs,
               std::vector<c10::optional<torch::Tensor>> masters, double lr, double b1,
               double b2, double eps, double wd, int64_t step);
std::vector<torch::Tensor> adam_prepare(std::vector<torch::Tensor> ps,
                                        std::vector<torch::Tensor> gs,
                                        std::vector<torch::Tensor> ms,
                                        std::vector<torch::Tensor> vs,
                                        std::vector<c10::optional<torch::Tensor>> masters);
void adam_apply(torch::Tensor desc, int64_t nchunks, torch::Tensor lr_dev,
                torch::Tensor step_dev, double b1, double b2, double eps, double wd);
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                    bool causal, double scale);
void attn_bwd_p(torch::Tensor s, torch::Tensor lse, bool causal, double scale);
torch::Tensor attn_d(torch::Tensor dO, torch::Tensor O);
void attn_ds(torch::Tensor dp, torch::Tensor p, torch::Tensor D, double scale);
void attn_bwd_fused(torch::Tensor q, torch::Tensor k, torch::Tensor v, torch::Tensor o,
                    torch::Tensor dO, torch::Tensor lse, torch::Tensor dq, torch::Tensor dk,
                    torch::Tensor dv, bool causal, double scale);
torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B);
torch::Tensor mfma_probe32(torch::Tensor A, torch::Tensor B);
bool gemm_supported(int64_t M, int64_t N, int64_t K, std::string dtype);
std::vector<torch::Tensor> l2norm_fwd(torch::Tensor x);
torch::Tensor l2norm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor rinv);
std::vector<torch::Tensor> xent_rows_fwd(torch::Tensor logits, torch::Tensor labels);
torch::Tensor xent_rows_bwd(torch::Tensor logits, torch::Tensor labels, torch::Tensor lse,
                            double gscale);
std::vector<torch::Tensor> sigmoid_loss_ew(torch::Tensor logits, int64_t diag0);
bool gemm8p_supported(int64_t M, int64_t N, int64_t K);
bool gemm_tn8p_supported(int64_t M, int64_t N, int64_t K);
torch::Tensor gemm_tn_8p(torch::Tensor dz, torch::Tensor x);
std::vector<torch::Tensor> gemm_tn_8p_db(torch::Tensor dz, torch::Tensor x);
torch::Tensor gemm_nt_8p_gradact(torch::Tensor dy, torch::Tensor wt, torch::Tensor z, std::string act);
torch::Tensor tr16_probe(torch::Tensor src, int64_t mode);
std::vector<torch::Tensor> linear_fwd(torch::Tensor x, torch::Tensor w,
                                      c10::optional<torch::Tensor> bias, std::string act,
                                      c10::optional<torch::Tensor> residual, bool save_z);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("layernorm_fwd", &layernorm_fwd, "LayerNorm forward (K3)");
  m.def("layernorm_bwd", &layernorm_bwd, "LayerNorm backward (K3/K15), optional fused addend",
        py::arg("dy"), py::arg("x"), py::arg("w"), py::arg("mean"), py::arg("rstd"),
        py::arg("addend") = py::none());
  m.def("bias_act_fwd", &bias_act_fwd, "fused bias+activation(+residual) forward");
  m.def("bias_act_fwd_fp8", &bias_act_fwd_fp8, "bias+act forward with fused e4m3 emit");
  m.def("gemm_nt_8p_gradact_fp8", &gemm_nt_8p_gradact_fp8,
        "gradact dX GEMM with fused e4m3 emit of dz");
  m.def("layernorm_fwd_fp8", &layernorm_fwd_fp8, "LayerNorm forward with fused e4m3 emit");
  m.def("act_bwd", &act_bwd, "activation backward: dy * act'(z)");
  m.def("colsum", &colsum, "column sum -> fp32 (bias gradient, K15)");
  m.def("im2col_patch", &im2col_patch, "patch-embed unfold (K1)");
  m.def("cls_pos_fwd", &cls_pos_fwd, "fused CLS concat + pos-emb add (K2)");
  m.def("embed_pos_fwd", &embed_pos_fwd, "fused token-embed gather + pos-emb add (K10)");
  m.def("col2im_patch", &col2im_patch, "patch-embed fold backward (K1/K15)");
  m.def("adam_step", &adam_step, "fused multi-tensor Adam (K14)");
  m.def("adam_prepare", &adam_prepare, "build device chunk descriptors once (K14)");
  m.def("adam_apply", &adam_apply, "graph-capturable fused Adam update (K14)");
  m.def("attn_fwd", &attn_fwd, "flash attention forward, head_dim 64 (K5)");
  m.def("attn_bwd_p", &attn_bwd_p, "attention bwd: S -> P in place (K15)");
  m.def("attn_d", &attn_d, "attention bwd: rowsum(dO*O) (K15)");
  m.def("attn_ds", &attn_ds, "attention bwd: dP -> dS in place (K15)");
  m.def("attn_bwd_fused", &attn_bwd_fused, "fused flash attention backward (K15)");
  m.def("mfma_probe", &mfma_probe, "MFMA 16x16x32 layout probe");
  m.def("mfma_probe32", &mfma_probe32, "MFMA 32x32x16 layout probe");
  m.def("gemm_supported", &gemm_supported, "MFMA GEMM shape support check");
  m.def("gemm8p_supported", &gemm8p_supported, "8-phase 256-tile GEMM shape check");
  m.def("gemm_tn8p_supported", &gemm_tn8p_supported, "TN dW GEMM shape check");
  m.def("gemm_tn_8p", &gemm_tn_8p, "TN weight-grad GEMM, split-M, tr_b16 (K15)");
  m.def("gemm_tn_8p_db", &gemm_tn_8p_db, "split-M TN dW with fused bias-grad colsum");
  m.def("gemm_nt_8p_gradact", &gemm_nt_8p_gradact, "dX GEMM with fused act-backward epilogue");
  m.def("tr16_probe", &tr16_probe, "ds_read_b64_tr_b16 lane-mapping probe");
  m.def("l2norm_fwd", &l2norm_fwd, "row L2-normalize forward (K12)");
  m.def("l2norm_bwd", &l2norm_bwd, "row L2-normalize backward (K12)");
  m.def("xent_rows_fwd", &xent_rows_fwd, "fused softmax-CE forward (K13)");
  m.def("xent_rows_bwd", &xent_rows_bwd, "fused softmax-CE backward (K13)");
  m.def("sigmoid_loss_ew", &sigmoid_loss_ew, "SigLIP sigmoid loss + dLogits (K13)");
  m.def("linear_fwd", &linear_fwd, "MFMA GEMM + fused epilogue (K4/K6/K7/K8)");
}
