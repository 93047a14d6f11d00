// Python bindings for the gfx950 HIP op library.
#include <torch/extension.h>

#include <vector>

std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w,
                                      at::Tensor b, double eps);
std::vector<at::Tensor> layernorm_bwd(at::Tensor dy, at::Tensor x,
                                      at::Tensor w, at::Tensor mean,
                                      at::Tensor rstd);
at::Tensor geglu_fwd(at::Tensor x);
at::Tensor geglu_bwd(at::Tensor dy, at::Tensor x);
at::Tensor dist_buckets(at::Tensor coords, at::Tensor boundaries);
at::Tensor pcgemm(at::Tensor A, at::Tensor B, long Bb, long M, long N,
                  long K, long D,
                  long a_bs, long a_ms, long a_ks,
                  long b_bs, long b_ns, long b_ks, double alpha);
at::Tensor gatemul_fwd(at::Tensor x, at::Tensor g, long xs, long gs,
                       c10::optional<at::Tensor> rowmask);
std::vector<at::Tensor> gatemul_bwd(at::Tensor dy, at::Tensor x, at::Tensor g,
                                    long xs, long gs,
                                    c10::optional<at::Tensor> rowmask,
                                    c10::optional<at::Tensor> dx_out,
                                    c10::optional<at::Tensor> dg_out,
                                    long dxs, long dgs);
at::Tensor linear_fwd(at::Tensor x, at::Tensor W,
                      c10::optional<at::Tensor> bias,
                      c10::optional<at::Tensor> resid, long stage);
std::vector<at::Tensor> ff1_geglu_fwd(at::Tensor x, at::Tensor W,
                                      c10::optional<at::Tensor> bias,
                                      long stage, bool want_inter);
at::Tensor wgrad(at::Tensor dY, at::Tensor X);
at::Tensor ipa_core_fwd(at::Tensor q_s, at::Tensor k_s, at::Tensor v_s,
                        at::Tensor q_pg, at::Tensor k_pg, at::Tensor v_pg,
                        at::Tensor bias, at::Tensor pair, at::Tensor rot,
                        at::Tensor trans, at::Tensor point_w,
                        double scale_s, double scale_b, double scale_p,
                        double eps);
at::Tensor pairrep_fwd(at::Tensor left, at::Tensor right, at::Tensor emb,
                       at::Tensor rel);
std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 c10::optional<at::Tensor> bias,
                                 c10::optional<at::Tensor> mask,
                                 long bias_repeat, double scale,
                                 long q_repeat);
std::vector<at::Tensor> attn_bwd(at::Tensor dout, at::Tensor q, at::Tensor k,
                                 at::Tensor v, at::Tensor out, at::Tensor lse,
                                 c10::optional<at::Tensor> bias,
                                 c10::optional<at::Tensor> mask,
                                 long bias_repeat, double scale,
                                 bool need_dbias, long q_repeat,
                                 c10::optional<at::Tensor> dq_out,
                                 c10::optional<at::Tensor> dk_out,
                                 c10::optional<at::Tensor> dv_out);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("layernorm_fwd", &layernorm_fwd, "fused LayerNorm forward (gfx950)");
  m.def("layernorm_bwd", &layernorm_bwd, "fused LayerNorm backward (gfx950)");
  m.def("geglu_fwd", &geglu_fwd, "fused GEGLU forward (gfx950)");
  m.def("geglu_bwd", &geglu_bwd, "fused GEGLU backward (gfx950)");
  m.def("dist_buckets", &dist_buckets, "fused cdist+bucketize (gfx950)");
  m.def("pcgemm", &pcgemm,
        "per-channel batched GEMM, stride-parameterized (gfx950 MFMA)");
  m.def("gatemul_fwd", &gatemul_fwd, "fused x*sigmoid(g) forward (gfx950)",
        py::arg("x"), py::arg("g"), py::arg("xs"), py::arg("gs"),
        py::arg("rowmask") = c10::nullopt);
  m.def("gatemul_bwd", &gatemul_bwd, "fused x*sigmoid(g) backward (gfx950)",
        py::arg("dy"), py::arg("x"), py::arg("g"), py::arg("xs"),
        py::arg("gs"), py::arg("rowmask") = c10::nullopt,
        py::arg("dx_out") = c10::nullopt, py::arg("dg_out") = c10::nullopt,
        py::arg("dxs") = 0, py::arg("dgs") = 0);
  m.def("linear_fwd", &linear_fwd,
        "tall-M small-K linear GEMM, bias/residual epilogue (gfx950 MFMA)",
        py::arg("x"), py::arg("W"), py::arg("bias") = c10::nullopt,
        py::arg("resid") = c10::nullopt, py::arg("stage") = -1);
  m.def("ff1_geglu_fwd", &ff1_geglu_fwd,
        "linear GEMM with fused GEGLU epilogue (gfx950 MFMA)",
        py::arg("x"), py::arg("W"), py::arg("bias") = c10::nullopt,
        py::arg("stage") = -1, py::arg("want_inter") = true);
  m.def("pairrep_fwd", &pairrep_fwd,
        "fused pair-rep build: outer sum + rel-pos embedding gather "
        "(gfx950, K13)", py::arg("left"), py::arg("right"), py::arg("emb"),
        py::arg("rel"));
  m.def("ipa_core_fwd", &ipa_core_fwd,
        "fused fp32 IPA attention core, inference path (gfx950, K7)");
  m.def("wgrad", &wgrad,
        "split-K weight-gradient GEMM dY^T @ X (gfx950 MFMA, fp32 out)",
        py::arg("dY"), py::arg("X"));
  m.def("attn_fwd", &attn_fwd, "fused flash attention forward (gfx950)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("bias"),
        py::arg("mask"), py::arg("bias_repeat"), py::arg("scale"),
        py::arg("q_repeat") = 1);
  m.def("attn_bwd", &attn_bwd, "fused flash attention backward (gfx950)",
        py::arg("dout"), py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("out"), py::arg("lse"), py::arg("bias"), py::arg("mask"),
        py::arg("bias_repeat"), py::arg("scale"), py::arg("need_dbias"),
        py::arg("q_repeat") = 1,
        py::arg("dq_out") = c10::nullopt, py::arg("dk_out") = c10::nullopt,
        py::arg("dv_out") = c10::nullopt);
}
