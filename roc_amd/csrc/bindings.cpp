// Python bindings for the roc_amd CDNA4 kernel library (roc_amd._C).
#include <torch/extension.h>

void spmm(torch::Tensor out, torch::Tensor x, torch::Tensor rowptr,
          torch::Tensor colidx, c10::optional<torch::Tensor> deg_dst,
          c10::optional<torch::Tensor> deg_src,
          c10::optional<torch::Tensor> row_order, bool accumulate,
          int64_t col_base, int64_t ncols);
void rowscale(torch::Tensor out, torch::Tensor x, torch::Tensor scale);
void cast_rowscale(torch::Tensor out, torch::Tensor x,
                   c10::optional<torch::Tensor> scale);
void relu_fwd(torch::Tensor out, torch::Tensor x);
void sigmoid_fwd(torch::Tensor out, torch::Tensor x);
void relu_bwd(torch::Tensor dx, torch::Tensor dy, torch::Tensor y);
void sigmoid_bwd(torch::Tensor dx, torch::Tensor dy, torch::Tensor y);
void ewise_add(torch::Tensor out, torch::Tensor a, torch::Tensor b);
void ewise_mul(torch::Tensor out, torch::Tensor a, torch::Tensor b);
void dropout_fwd(torch::Tensor out, torch::Tensor x, double p, int64_t seed,
                 int64_t offset, c10::optional<torch::Tensor> counter);
void softmax_ce(torch::Tensor dl, torch::Tensor metrics, torch::Tensor logits,
                torch::Tensor labels, torch::Tensor mask, double grad_scale,
                int64_t num_classes);
void adam_step(torch::Tensor w, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, double alpha, double b1, double b2, double eps,
               double wd, c10::optional<torch::Tensor> step,
               double decay_rate, int64_t decay_steps);
void gemm_rr(torch::Tensor C, torch::Tensor A, torch::Tensor Bt, bool relu,
             c10::optional<torch::Tensor> row_scale);
void gemm_atb(torch::Tensor C, torch::Tensor A, torch::Tensor B);
void spmm_refresh_knobs();
void spmm_edge(torch::Tensor out, torch::Tensor x, torch::Tensor rowptr,
               torch::Tensor colidx, torch::Tensor edge_val,
               c10::optional<torch::Tensor> deg_dst,
               c10::optional<torch::Tensor> row_order, bool accumulate);
void edge_dot(torch::Tensor dw, torch::Tensor dy, torch::Tensor x,
              torch::Tensor rowptr, torch::Tensor colidx);
void edge_softmax_fwd(torch::Tensor alpha, torch::Tensor s,
                      torch::Tensor rowptr);
void edge_softmax_bwd(torch::Tensor ds, torch::Tensor dalpha,
                      torch::Tensor alpha, torch::Tensor rowptr);
void att_softmax_fwd(torch::Tensor alpha, torch::Tensor s_src,
                     torch::Tensor s_dst, torch::Tensor rowptr,
                     torch::Tensor colidx, double slope);
void att_softmax_bwd(torch::Tensor dsrc, torch::Tensor dsdst,
                     torch::Tensor dalpha, torch::Tensor alpha,
                     torch::Tensor s_src, torch::Tensor s_dst,
                     torch::Tensor rowptr, torch::Tensor colidx,
                     double slope);
void register_graph_cpu(pybind11::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "roc_amd hand-written CDNA4 (gfx950) kernels";
  m.def("spmm", &spmm, "CSR SpMM neighbor aggregation (fused deg-norm)",
        pybind11::arg("out"), pybind11::arg("x"), pybind11::arg("rowptr"),
        pybind11::arg("colidx"), pybind11::arg("deg_dst") = pybind11::none(),
        pybind11::arg("deg_src") = pybind11::none(),
        pybind11::arg("row_order") = pybind11::none(),
        pybind11::arg("accumulate") = false,
        pybind11::arg("col_base") = 0, pybind11::arg("ncols") = 0);
  m.def("rowscale", &rowscale);
  m.def("cast_rowscale", &cast_rowscale,
        "fp32 -> bf16 cast with optional row scale (strip epilogue)",
        pybind11::arg("out"), pybind11::arg("x"),
        pybind11::arg("scale") = pybind11::none());
  m.def("relu_fwd", &relu_fwd);
  m.def("sigmoid_fwd", &sigmoid_fwd);
  m.def("relu_bwd", &relu_bwd);
  m.def("sigmoid_bwd", &sigmoid_bwd);
  m.def("ewise_add", &ewise_add);
  m.def("ewise_mul", &ewise_mul);
  m.def("dropout_fwd", &dropout_fwd, pybind11::arg("out"), pybind11::arg("x"),
        pybind11::arg("p"), pybind11::arg("seed"), pybind11::arg("offset"),
        pybind11::arg("counter") = pybind11::none());
  m.def("softmax_ce", &softmax_ce, pybind11::arg("dl"), pybind11::arg("metrics"),
        pybind11::arg("logits"), pybind11::arg("labels"), pybind11::arg("mask"),
        pybind11::arg("grad_scale"), pybind11::arg("num_classes") = -1);
  m.def("adam_step", &adam_step, pybind11::arg("w"), pybind11::arg("g"),
        pybind11::arg("m"), pybind11::arg("v"), pybind11::arg("alpha"),
        pybind11::arg("b1"), pybind11::arg("b2"), pybind11::arg("eps"),
        pybind11::arg("wd"), pybind11::arg("step") = pybind11::none(),
        pybind11::arg("decay_rate") = 1.0,
        pybind11::arg("decay_steps") = 100);
  m.def("gemm_rr", &gemm_rr, "C = A @ Bt^T (bf16 MFMA, fused relu/rowscale)",
        pybind11::arg("C"), pybind11::arg("A"), pybind11::arg("Bt"),
        pybind11::arg("relu"), pybind11::arg("row_scale") = pybind11::none());
  m.def("gemm_atb", &gemm_atb, "C += A^T @ B (fp32 split-K accumulate)");
  m.def("spmm_refresh_knobs", &spmm_refresh_knobs,
        "re-read ROC_SPMM_* env knobs (A/B harness only)");
  m.def("spmm_edge", &spmm_edge,
        "per-edge-weighted CSR SpMM (edge-tensor consumer)",
        pybind11::arg("out"), pybind11::arg("x"), pybind11::arg("rowptr"),
        pybind11::arg("colidx"), pybind11::arg("edge_val"),
        pybind11::arg("deg_dst") = pybind11::none(),
        pybind11::arg("row_order") = pybind11::none(),
        pybind11::arg("accumulate") = false);
  m.def("edge_dot", &edge_dot,
        "dw[e] = <dy[row_e], x[col_e]> (edge-value gradient)");
  m.def("edge_softmax_fwd", &edge_softmax_fwd,
        "per-row segment softmax over edge scores (GAT attention)");
  m.def("edge_softmax_bwd", &edge_softmax_bwd);
  m.def("att_softmax_fwd", &att_softmax_fwd,
        "fused GAT attention: lrelu(s_src[col]+s_dst[row]) -> softmax");
  m.def("att_softmax_bwd", &att_softmax_bwd);
  register_graph_cpu(m);
}
