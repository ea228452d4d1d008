// Python bindings for the ravnest_amd CDNA4 kernel library (_C).
#include <torch/extension.h>

#include <vector>

std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w,
                                      at::Tensor b, double eps);
std::vector<at::Tensor> layernorm_bwd(at::Tensor dy, at::Tensor x,
                                      at::Tensor w, at::Tensor mean,
                                      at::Tensor rstd);
std::vector<at::Tensor> layernorm_add_fwd(at::Tensor a, at::Tensor b,
                                          at::Tensor w, at::Tensor bias,
                                          double eps);
at::Tensor bias_gelu_fwd(at::Tensor x, at::Tensor bias);
at::Tensor bias_gelu_bwd(at::Tensor dy, at::Tensor x, at::Tensor bias);
at::Tensor colsum_bf16(at::Tensor x);
std::vector<at::Tensor> bias_gelu_bwd_db(at::Tensor dy, at::Tensor x,
                                         at::Tensor bias);
std::vector<at::Tensor> batchnorm_fwd(at::Tensor x, at::Tensor w,
                                      at::Tensor b, at::Tensor running_mean,
                                      at::Tensor running_var, bool training,
                                      double momentum, double eps);
std::vector<at::Tensor> batchnorm_bwd(at::Tensor dy, at::Tensor x,
                                      at::Tensor w, at::Tensor mean,
                                      at::Tensor rstd);
at::Tensor conv2d_fwd(at::Tensor x, at::Tensor w,
                      c10::optional<at::Tensor> bias, long st, long pad,
                      bool relu);
at::Tensor conv2d_dgrad(at::Tensor dy, at::Tensor w, long N, long H,
                        long W, long st, long pad);
at::Tensor conv2d_wgrad(at::Tensor dy, at::Tensor x, long R, long S,
                        long st, long pad);
std::vector<at::Tensor> mx_quant(at::Tensor x);
std::vector<at::Tensor> gemm_nt_bf16(at::Tensor A2, at::Tensor B,
                                     c10::optional<at::Tensor> bias,
                                     int64_t epi);
at::Tensor gemm_wgrad_bf16(at::Tensor dy, at::Tensor x);
at::Tensor mx_gemm(at::Tensor x, at::Tensor xs, at::Tensor w,
                   at::Tensor ws);
at::Tensor mx_gemm2(at::Tensor xq, at::Tensor xs, at::Tensor wq,
                    at::Tensor ws);
at::Tensor mx_scale_probe(at::Tensor a, at::Tensor b, at::Tensor sa,
                          at::Tensor sb);
at::Tensor softmax_fwd(at::Tensor x);
at::Tensor softmax_bwd(at::Tensor dy, at::Tensor y);
std::vector<at::Tensor> maxpool2d_fwd(at::Tensor x, long K, long S, long P);
at::Tensor maxpool2d_bwd(at::Tensor dy, at::Tensor idx, long H, long W,
                         long K, long S, long P);
at::Tensor avgpool2d_fwd(at::Tensor x, long K, long S, long P,
                         bool include_pad);
at::Tensor avgpool2d_bwd(at::Tensor dy, long H, long W, long K, long S,
                         long P, bool include_pad);
at::Tensor global_avgpool_fwd(at::Tensor x);
at::Tensor global_avgpool_bwd(at::Tensor dy, long H, long W);
std::vector<at::Tensor> emb2_ln_fwd(at::Tensor ids, at::Tensor word,
                                    at::Tensor pos, at::Tensor w,
                                    at::Tensor b, double eps);
at::Tensor emb2_add_fwd(at::Tensor ids, at::Tensor word, at::Tensor pos);
std::vector<at::Tensor> emb2_bwd(at::Tensor dx, at::Tensor ids, long V,
                                 long P);
std::vector<at::Tensor> dropout_fwd(at::Tensor x, double p, int64_t seed,
                                    c10::optional<at::Tensor> seed_buf);
at::Tensor dropout_bwd(at::Tensor dy, at::Tensor mask, double p);
std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor targets,
                               int64_t ignore_index);
at::Tensor ce_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse,
                  at::Tensor gscale, int64_t ignore_index);
std::tuple<at::Tensor, long, long> adam_build_table(
    std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
    std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
    std::vector<at::Tensor> masters);
void fused_adam_graph(at::Tensor table, long nchunks, long esize,
                      at::Tensor lr_buf, double b1, double b2, double eps,
                      double wd, at::Tensor step_buf);
void fused_adam(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
                std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                std::vector<at::Tensor> masters, double lr, double b1,
                double b2, double eps, double wd, double bc1, double bc2);
void fused_sgd(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
               std::vector<at::Tensor> bufs, std::vector<at::Tensor> masters,
               double lr, double momentum, double wd, bool nesterov);
void fused_lamb(std::vector<at::Tensor> ps, std::vector<at::Tensor> gs,
                std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                std::vector<at::Tensor> masters, double lr, double b1,
                double b2, double eps, double wd, double bc1, double bc2,
                double clamp_trust);
void fused_copy(std::vector<at::Tensor> srcs, std::vector<at::Tensor> dsts);
std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor mask, bool causal, double scale);
std::vector<at::Tensor> attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 at::Tensor o, at::Tensor dout,
                                 at::Tensor lse, at::Tensor mask, bool causal,
                                 double scale);
std::vector<at::Tensor> attn_fwd_qkv(at::Tensor qkv, at::Tensor mask,
                                     bool causal, double scale,
                                     double pdrop, int64_t dseed,
                                     c10::optional<at::Tensor> seed_buf);
at::Tensor attn_bwd_qkv(at::Tensor qkv, at::Tensor o, at::Tensor dout,
                        at::Tensor lse, at::Tensor mask, bool causal,
                        double scale, double pdrop,
                        c10::optional<at::Tensor> mbits);
at::Tensor mfma_probe(at::Tensor a, at::Tensor b);
at::Tensor tr16_probe();
at::Tensor mfma_mx_probe(at::Tensor a, at::Tensor b, int64_t sa, int64_t sb);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("layernorm_fwd", &layernorm_fwd, "fused LayerNorm fwd (CDNA4)");
  m.def("layernorm_bwd", &layernorm_bwd, "fused LayerNorm bwd (CDNA4)");
  m.def("layernorm_add_fwd", &layernorm_add_fwd,
        "fused residual-add + LayerNorm fwd");
  m.def("bias_gelu_fwd", &bias_gelu_fwd, "fused bias+GELU fwd");
  m.def("bias_gelu_bwd", &bias_gelu_bwd, "fused bias+GELU bwd");
  m.def("colsum_bf16", &colsum_bf16, "column sum bf16 -> fp32");
  m.def("bias_gelu_bwd_db", &bias_gelu_bwd_db,
        "fused bias+GELU bwd with in-pass bias-grad column reduction");
  m.def("batchnorm_fwd", &batchnorm_fwd, "fused BatchNorm2d fwd (NCHW)");
  m.def("batchnorm_bwd", &batchnorm_bwd, "fused BatchNorm2d bwd (NCHW)");
  m.def("dropout_fwd", &dropout_fwd, "philox dropout fwd (replayable)",
        py::arg("x"), py::arg("p"), py::arg("seed"),
        py::arg("seed_buf") = py::none());
  m.def("dropout_bwd", &dropout_bwd, "dropout bwd");
  m.def("conv2d_fwd", &conv2d_fwd,
        "implicit-GEMM MFMA conv fwd (bf16 NCHW)", py::arg("x"),
        py::arg("w"), py::arg("bias") = py::none(), py::arg("st") = 1,
        py::arg("pad") = 0, py::arg("relu") = false);
  m.def("conv2d_dgrad", &conv2d_dgrad, "implicit-GEMM MFMA conv dgrad");
  m.def("conv2d_wgrad", &conv2d_wgrad, "implicit-GEMM MFMA conv wgrad");
  m.def("gemm_nt_bf16", &gemm_nt_bf16,
        "bf16 MFMA GEMM x @ W^T (256^2 8-phase, fused bias/GELU epilogue)",
        py::arg("a"), py::arg("b"), py::arg("bias") = py::none(),
        py::arg("epi") = 0);
  m.def("gemm_wgrad_bf16", &gemm_wgrad_bf16,
        "split-K wgrad dW = dy^T @ x (tr16 transpose reads, fp32 atomics)");
  m.def("mx_quant", &mx_quant,
        "bf16 -> MX fp8 (e4m3 + per-32 e8m0 scales)");
  m.def("mx_gemm", &mx_gemm, "MX fp8 GEMM: x @ W^T, 32x32x64 scaled MFMA");
  m.def("mx_gemm2", &mx_gemm2,
        "LDS-staged MX fp8 GEMM (256^2 4-phase, scaled 32x32x64 MFMA)");
  m.def("mx_scale_probe", &mx_scale_probe,
        "per-lane scale-byte semantics probe");
  m.def("softmax_fwd", &softmax_fwd, "standalone softmax fwd (last dim)");
  m.def("softmax_bwd", &softmax_bwd, "standalone softmax bwd");
  m.def("maxpool2d_fwd", &maxpool2d_fwd, "MaxPool2d fwd (+argmax idx)");
  m.def("maxpool2d_bwd", &maxpool2d_bwd, "MaxPool2d bwd (gather)");
  m.def("avgpool2d_fwd", &avgpool2d_fwd, "AvgPool2d fwd");
  m.def("avgpool2d_bwd", &avgpool2d_bwd, "AvgPool2d bwd");
  m.def("global_avgpool_fwd", &global_avgpool_fwd,
        "global (adaptive 1x1) avg pool fwd");
  m.def("global_avgpool_bwd", &global_avgpool_bwd, "global avg pool bwd");
  m.def("emb2_ln_fwd", &emb2_ln_fwd,
        "fused word+pos embedding gather + LayerNorm fwd");
  m.def("emb2_add_fwd", &emb2_add_fwd, "fused word+pos embedding gather");
  m.def("emb2_bwd", &emb2_bwd,
        "embedding bwd: dword scatter-add + dpos batch reduction");
  m.def("ce_fwd", &ce_fwd, "fused softmax cross-entropy fwd");
  m.def("ce_bwd", &ce_bwd, "fused softmax cross-entropy bwd");
  m.def("fused_adam", &fused_adam, "multi-tensor Adam");
  m.def("adam_build_table", &adam_build_table,
        "prebuild the device chunk table (graph-capturable Adam)");
  m.def("fused_adam_graph", &fused_adam_graph,
        "multi-tensor Adam with device lr/step buffers (hipGraph mode)");
  m.def("fused_sgd", &fused_sgd, "multi-tensor SGD+momentum");
  m.def("fused_lamb", &fused_lamb, "multi-tensor LAMB");
  m.def("fused_copy", &fused_copy, "multi-tensor device copy");
  m.def("attn_fwd", &attn_fwd, "flash attention fwd (bf16, MFMA)");
  m.def("attn_bwd", &attn_bwd, "flash attention bwd (bf16, MFMA, D=64)");
  m.def("attn_fwd_qkv", &attn_fwd_qkv,
        py::arg("qkv"), py::arg("mask"), py::arg("causal"),
        py::arg("scale"), py::arg("pdrop") = 0.0, py::arg("dseed") = 0,
        py::arg("seed_buf") = py::none(),
        "flash attention fwd, packed (B,S,3,H,D) qkv -> (B,S,H,D)");
  m.def("attn_bwd_qkv", &attn_bwd_qkv,
        py::arg("qkv"), py::arg("o"), py::arg("dout"), py::arg("lse"),
        py::arg("mask"), py::arg("causal"), py::arg("scale"),
        py::arg("pdrop") = 0.0, py::arg("mbits") = py::none(),
        "flash attention bwd, packed qkv -> dqkv");
  m.def("mfma_probe", &mfma_probe, "32x32x16 bf16 MFMA layout probe");
  m.def("tr16_probe", &tr16_probe, "ds_read_b64_tr_b16 mapping probe");
  m.def("mfma_mx_probe", &mfma_mx_probe,
        "32x32x64 MX-scaled fp8 MFMA layout probe");
}
