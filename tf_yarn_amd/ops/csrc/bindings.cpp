// Python bindings for the tf_yarn_amd MI355X kernel library.
#include <torch/extension.h>
#include <vector>

// fused_optimizers.hip
void fused_sgd(torch::Tensor param, torch::Tensor grad,
               c10::optional<torch::Tensor> momentum_buf,
               c10::optional<torch::Tensor> param_bf16,
               double lr, double momentum, double dampening,
               double weight_decay, bool nesterov, bool first_step,
               double grad_scale);
void fused_sgd_mt(std::vector<torch::Tensor> params,
                  std::vector<torch::Tensor> grads,
                  std::vector<torch::Tensor> momentum_bufs,
                  std::vector<torch::Tensor> params_bf16,
                  double lr, double momentum, double dampening,
                  double weight_decay, bool nesterov, bool first_step,
                  double grad_scale);
void fused_adam(torch::Tensor param, torch::Tensor grad,
                torch::Tensor exp_avg, torch::Tensor exp_avg_sq,
                c10::optional<torch::Tensor> param_bf16,
                double lr, double beta1, double beta2, double eps,
                double weight_decay, bool adamw, int64_t step,
                double grad_scale);
void fused_adagrad(torch::Tensor param, torch::Tensor grad,
                   torch::Tensor state_sum, double lr, double eps,
                   double weight_decay, double grad_scale);
void fused_adadelta(torch::Tensor param, torch::Tensor grad,
                    torch::Tensor square_avg, torch::Tensor acc_delta,
                    double lr, double rho, double eps, double weight_decay,
                    double grad_scale);

// embedding.hip
torch::Tensor emb_fwd(torch::Tensor table, torch::Tensor ids, bool out_bf16);
void emb_bwd_sgd(torch::Tensor table, torch::Tensor ids, torch::Tensor grad,
                 double lr, double scale);
void emb_bwd_sgd_sorted(torch::Tensor table, torch::Tensor sorted_ids,
                        torch::Tensor grad, double lr, double scale);
void emb_bwd_dense(torch::Tensor grad_table, torch::Tensor ids,
                   torch::Tensor grad, double scale);
void emb_bwd_sgd_fused_wide(torch::Tensor table, torch::Tensor wide_table,
                            torch::Tensor ids, torch::Tensor grad,
                            torch::Tensor gw, double lr, double scale);

torch::Tensor emb_gather_sum(torch::Tensor table, torch::Tensor ids,
                             int64_t batch, bool out_bf16);
void emb_fwd_into(torch::Tensor table, torch::Tensor ids,
                  torch::Tensor out, int64_t col_offset);
void emb_scatter_sum(torch::Tensor table, torch::Tensor ids,
                     torch::Tensor grad, double alpha);

// binned_scatter.hip
std::vector<torch::Tensor> binned_permutation(torch::Tensor ids,
                                              int64_t n_rows,
                                              int64_t region_bits);
void emb_bwd_sgd_binned(torch::Tensor table, torch::Tensor ids,
                        torch::Tensor grad, double lr, double scale,
                        torch::Tensor order, torch::Tensor starts);
void emb_scatter_sum_binned(torch::Tensor table, torch::Tensor ids,
                            torch::Tensor grad, int64_t g_div,
                            double alpha, torch::Tensor order,
                            torch::Tensor starts);

// elementwise.hip
torch::Tensor bias_relu_fwd(torch::Tensor x, torch::Tensor bias);
torch::Tensor bias_relu_bwd(torch::Tensor dy, torch::Tensor y);
std::vector<torch::Tensor> bias_relu_bwd_db(torch::Tensor dy,
                                            torch::Tensor y,
                                            bool dbias_bf16);
torch::Tensor reduce_splitk(torch::Tensor part, bool out_bf16);

// lt_linear.hip
torch::Tensor lt_linear(torch::Tensor x, torch::Tensor w,
                        c10::optional<torch::Tensor> bias, bool relu);
torch::Tensor col_reduce_dot(torch::Tensor x, torch::Tensor dy);
torch::Tensor row_dot(torch::Tensor x, torch::Tensor w,
                      c10::optional<torch::Tensor> bias);
std::vector<torch::Tensor> bce_head_fwd(torch::Tensor deep,
                                        torch::Tensor wide,
                                        torch::Tensor dhead,
                                        torch::Tensor labels);
torch::Tensor bce_head_bwd(torch::Tensor sig, torch::Tensor labels,
                           torch::Tensor g);

// wgrad.hip
torch::Tensor wgrad_nt(torch::Tensor dy, torch::Tensor x, int64_t splitk,
                       bool out_bf16);
torch::Tensor wgrad_nt128(torch::Tensor dy, torch::Tensor x,
                          int64_t splitk, bool out_bf16);
torch::Tensor wgrad_nt256(torch::Tensor dy, torch::Tensor x,
                          int64_t splitk, bool out_bf16);
void convert_scaled(torch::Tensor src, torch::Tensor dst, double scale);

// gemm_bt.hip
torch::Tensor gemm_bt(torch::Tensor a, torch::Tensor b,
                      c10::optional<torch::Tensor> bias, bool relu);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "tf_yarn_amd MI355X (gfx950) HIP kernels";
  m.def("fused_sgd", &fused_sgd, "Fused SGD(+momentum) step");
  m.def("fused_sgd_mt", &fused_sgd_mt,
        "Multi-tensor fused SGD (one launch per <=24 tensors)");
  m.def("fused_adam", &fused_adam, "Fused Adam/AdamW step");
  m.def("fused_adagrad", &fused_adagrad, "Fused Adagrad step");
  m.def("fused_adadelta", &fused_adadelta, "Fused Adadelta step");
  m.def("emb_fwd", &emb_fwd, "Fused multi-table embedding gather");
  m.def("emb_bwd_sgd", &emb_bwd_sgd,
        "Fused sparse embedding grad scatter + SGD update");
  m.def("emb_bwd_sgd_sorted", &emb_bwd_sgd_sorted,
        "Atomic-free segmented scatter+SGD over SORTED ids");
  m.def("emb_bwd_sgd_fused_wide", &emb_bwd_sgd_fused_wide,
        "Fused deep scatter+SGD + wide scalar scatter (shared ids)");
  m.def("emb_bwd_dense", &emb_bwd_dense,
        "Sparse embedding grad scatter into dense grad table");
  m.def("bias_relu_fwd", &bias_relu_fwd, "Fused bias+ReLU forward");
  m.def("bias_relu_bwd", &bias_relu_bwd, "Fused ReLU backward");
  m.def("lt_linear", &lt_linear,
        "hipBLASLt linear with fused bias(+ReLU) epilogue",
        py::arg("x"), py::arg("w"), py::arg("bias") = py::none(),
        py::arg("relu") = false);
  m.def("reduce_splitk", &reduce_splitk,
        "Split-K partial reduction with fused output cast",
        py::arg("part"), py::arg("out_bf16") = false);
  m.def("wgrad_nt", &wgrad_nt,
        "Split-K MFMA weight gradient: dW = dy^T @ x (bf16 in)",
        py::arg("dy"), py::arg("x"), py::arg("splitk"),
        py::arg("out_bf16") = false);
  m.def("wgrad_nt128", &wgrad_nt128,
        "Split-K MFMA weight gradient, 128x128 tiles + XOR-swizzled LDS",
        py::arg("dy"), py::arg("x"), py::arg("splitk"),
        py::arg("out_bf16") = false);
  m.def("wgrad_nt256", &wgrad_nt256,
        "Split-K MFMA weight gradient, 256x256 tiles (8 waves)",
        py::arg("dy"), py::arg("x"), py::arg("splitk"),
        py::arg("out_bf16") = false);
  m.def("col_reduce_dot", &col_reduce_dot,
        "dw[m] = sum_b dy[b] * x[b,m] (single-logit head wgrad)");
  m.def("row_dot", &row_dot,
        "y[b] = x[b].w + bias (single-logit head forward GEMV)");
  m.def("bce_head_fwd", &bce_head_fwd,
        "Fused 3-part logit sum + stable BCEWithLogits (loss, sigmoid)");
  m.def("bce_head_bwd", &bce_head_bwd,
        "dlogit = (sigmoid - label) * upstream grad");
  m.def("bias_relu_bwd_db", &bias_relu_bwd_db,
        "Fused ReLU backward + dbias reduction (returns [dx, dbias])",
        py::arg("dy"), py::arg("y"), py::arg("dbias_bf16") = false);
  m.def("emb_fwd_into", &emb_fwd_into,
        "Embedding gather into a slice of a larger 2D buffer");
  m.def("emb_gather_sum", &emb_gather_sum,
        "Wide-part gather-sum: out[b] = sum_f table[ids[b,f]]");
  m.def("binned_permutation", &binned_permutation,
        "Region-binned permutation of sparse update indices");
  m.def("emb_bwd_sgd_binned", &emb_bwd_sgd_binned,
        "Binned LDS-dedup scatter+SGD (deep tables, dim=16)");
  m.def("emb_scatter_sum_binned", &emb_scatter_sum_binned,
        "Binned LDS-dedup scatter-add (scalar wide tables)");
  m.def("emb_scatter_sum", &emb_scatter_sum,
        "Wide-part scatter: table[ids[b,f]] += alpha * g[b]");
  m.def("convert_scaled", &convert_scaled, "Scaled bf16<->fp32 convert");
  m.def("gemm_bt", &gemm_bt,
        "C = A @ B^T (both K-major bf16) with fused bias+ReLU epilogue");
}
