// parallax_amd native extension: pybind bindings for the hand-written
// CDNA4 (gfx950) kernels.  Built in-tree by parallax_amd/ops/build.py with
// PYTORCH_ROCM_ARCH=gfx950.

#include <torch/extension.h>

#include <vector>

// sparse_kernels.hip
at::Tensor rows_gather(const at::Tensor& weight, const at::Tensor& ids);
at::Tensor segment_sum_rows(const at::Tensor& values, const at::Tensor& order,
                            const at::Tensor& seg_starts, bool average);

// optim_kernels.hip
void multi_tensor_sgd(std::vector<at::Tensor> params,
                      std::vector<at::Tensor> grads, double lr, double wd,
                      std::vector<at::Tensor> masters);
void multi_tensor_momentum(std::vector<at::Tensor> params,
                           std::vector<at::Tensor> grads,
                           std::vector<at::Tensor> bufs, double lr,
                           double mom, double wd, bool nesterov,
                           std::vector<at::Tensor> masters);
void multi_tensor_adagrad(std::vector<at::Tensor> params,
                          std::vector<at::Tensor> grads,
                          std::vector<at::Tensor> accums, double lr,
                          double eps, std::vector<at::Tensor> masters);
void multi_tensor_adam(std::vector<at::Tensor> params,
                       std::vector<at::Tensor> grads,
                       std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                       std::vector<double> steps, double lr, double b1,
                       double b2, double eps, double wd,
                       std::vector<at::Tensor> masters);
void multi_tensor_adam_dev(std::vector<at::Tensor> params,
                           std::vector<at::Tensor> grads,
                           std::vector<at::Tensor> ms,
                           std::vector<at::Tensor> vs, at::Tensor step_dev,
                           double lr, double b1, double b2, double eps,
                           double wd, std::vector<at::Tensor> masters);
void multi_tensor_rmsprop(std::vector<at::Tensor> params,
                          std::vector<at::Tensor> grads,
                          std::vector<at::Tensor> sqs,
                          std::vector<at::Tensor> bufs, double lr,
                          double decay, double mom, double eps,
                          std::vector<at::Tensor> masters);
void multi_tensor_ema(std::vector<at::Tensor> shadows,
                      std::vector<at::Tensor> values, double decay);
void multi_tensor_zero(std::vector<at::Tensor> tensors);
at::Tensor multi_tensor_l2norm(std::vector<at::Tensor> tensors);
void multi_tensor_scale_by_norm(std::vector<at::Tensor> tensors,
                                at::Tensor norm, double clip);
void sparse_adagrad(at::Tensor param, at::Tensor accum, at::Tensor ids,
                    at::Tensor grad_rows, double lr, double eps,
                    c10::optional<at::Tensor> nvalid);
void sparse_sgd(at::Tensor param, at::Tensor ids, at::Tensor grad_rows,
                double lr, c10::optional<at::Tensor> nvalid);
void multi_tensor_pack(std::vector<at::Tensor> tensors, at::Tensor flat);
void multi_tensor_unpack(std::vector<at::Tensor> tensors, at::Tensor flat,
                         double scale);

// gru_cell.hip
std::vector<at::Tensor> gru_gates_fwd(const at::Tensor& xg,
                                      const at::Tensor& hg,
                                      const at::Tensor& h_prev);
std::vector<at::Tensor> gru_gates_bwd(
    const at::Tensor& dh, const at::Tensor& h_prev, const at::Tensor& r_s,
    const at::Tensor& z_s, const at::Tensor& n_t, const at::Tensor& hgn_s);

// batchnorm.hip
std::vector<at::Tensor> batchnorm_fwd(const at::Tensor& x2d,
                                      const at::Tensor& gamma,
                                      const at::Tensor& beta,
                                      c10::optional<at::Tensor> run_mean,
                                      c10::optional<at::Tensor> run_var,
                                      double momentum, double eps,
                                      bool relu,
                                      c10::optional<at::Tensor> addend);
std::vector<at::Tensor> batchnorm_bwd(const at::Tensor& dy2d,
                                      const at::Tensor& x2d,
                                      const at::Tensor& y2d,
                                      const at::Tensor& mean,
                                      const at::Tensor& invstd,
                                      const at::Tensor& gamma, bool relu,
                                      bool want_dxid);

// pool.hip
std::vector<at::Tensor> maxpool3x3s2_fwd(const at::Tensor& x2d, long N,
                                         long H, long W, long C);
at::Tensor maxpool3x3s2_bwd(const at::Tensor& dy, const at::Tensor& arg,
                            long N, long H, long W, long C);
at::Tensor gavgpool_fwd(const at::Tensor& x2d, long N, long HW, long C);
at::Tensor gavgpool_bwd(const at::Tensor& dy, long N, long HW, long C);

// reduce.hip
at::Tensor col_sum(const at::Tensor& x);

// gemm.hip
at::Tensor pxgemm_bt(const at::Tensor& a, const at::Tensor& bt);
at::Tensor pxgemm8_bt(const at::Tensor& a, const at::Tensor& bt);
at::Tensor pxgemm8_bt_v(const at::Tensor& a, const at::Tensor& bt,
                        long variant);
at::Tensor pxgemm8_bt_add(const at::Tensor& a, const at::Tensor& bt,
                          const c10::optional<at::Tensor>& addend);
at::Tensor pxgemm_bt_add(const at::Tensor& a, const at::Tensor& bt,
                         const c10::optional<at::Tensor>& addend);
at::Tensor px_tall_bt(const at::Tensor& a, const at::Tensor& bt);
at::Tensor px_wgrad3x3(const at::Tensor& dy2d, const at::Tensor& xp,
                       long n_img, long oh, long ow, long cin,
                       long stride, long HP, long WP);
at::Tensor px_tall_conv3x3(const at::Tensor& xp, const at::Tensor& w2d,
                           long n_img, long oh, long ow, long cin,
                           long stride, long HP, long WP);
void px_skinny_bt(const at::Tensor& a, const at::Tensor& bt,
                  at::Tensor ws, at::Tensor cnt, at::Tensor out,
                  const c10::optional<at::Tensor>& addend);
at::Tensor pxconv3x3_fwd(const at::Tensor& xp, const at::Tensor& w2d,
                         long n_img, long oh, long ow, long cin,
                         long stride, long HP, long WP);

// attention.hip
at::Tensor px_attn_fwd(const at::Tensor& q, const at::Tensor& k,
                       const at::Tensor& v, bool causal);
std::vector<at::Tensor> px_attn_bwd(const at::Tensor& q,
                                    const at::Tensor& k,
                                    const at::Tensor& v,
                                    const at::Tensor& dout, bool causal);

// layernorm.hip
std::vector<at::Tensor> layernorm_fwd(const at::Tensor& x,
                                      const at::Tensor& w,
                                      const at::Tensor& b, double eps);
std::vector<at::Tensor> layernorm_bwd(const at::Tensor& dy,
                                      const at::Tensor& x,
                                      const at::Tensor& w,
                                      const at::Tensor& mean,
                                      const at::Tensor& rstd);

// softmax_ce.hip
std::vector<at::Tensor> softmax_ce_fwd(const at::Tensor& logits,
                                       const at::Tensor& labels);
at::Tensor softmax_ce_bwd(const at::Tensor& logits, const at::Tensor& labels,
                          const at::Tensor& lse, const at::Tensor& dloss);
std::vector<at::Tensor> sampled_ce_fwd(const at::Tensor& true_logits,
                                       const at::Tensor& sampled,
                                       const at::Tensor& row_corr,
                                       const at::Tensor& col_corr);
std::vector<at::Tensor> sampled_ce_bwd(const at::Tensor& true_logits,
                                       const at::Tensor& sampled,
                                       const at::Tensor& row_corr,
                                       const at::Tensor& col_corr,
                                       const at::Tensor& lse,
                                       const at::Tensor& dloss);

// lstm_cell.hip
std::vector<at::Tensor> lstm_gates_fwd(const at::Tensor& gates,
                                       const at::Tensor& c_prev,
                                       double forget_bias);
std::vector<at::Tensor> lstm_gates_bwd(
    const at::Tensor& dh, const at::Tensor& dc_in, const at::Tensor& c_prev,
    const at::Tensor& c_out, const at::Tensor& i_s, const at::Tensor& g_t,
    const at::Tensor& f_s, const at::Tensor& o_s);
void lstm_gates_fwd_out(const at::Tensor& gates, const at::Tensor& c_prev,
                        double forget_bias, at::Tensor h_out,
                        at::Tensor c_out, at::Tensor i_s, at::Tensor g_t,
                        at::Tensor f_s, at::Tensor o_s);
// lstm_fused.hip
void lstm_step_fwd_fused(const at::Tensor& hp, const at::Tensor& whT,
                         const at::Tensor& gx_t, const at::Tensor& c_prev,
                         double forget_bias, at::Tensor s_out,
                         at::Tensor c_out, at::Tensor i_s, at::Tensor g_t,
                         at::Tensor f_s, at::Tensor o_s);
void lstm_step_bwd_fused(const at::Tensor& dhp, const at::Tensor& wproj,
                         const at::Tensor& dc_in, const at::Tensor& c_prev,
                         const at::Tensor& c_out, const at::Tensor& i_s,
                         const at::Tensor& g_t, const at::Tensor& f_s,
                         const at::Tensor& o_s, at::Tensor dgates,
                         at::Tensor dc_prev);
void lstm_gates_bwd_out(const at::Tensor& dh, const at::Tensor& dc_in,
                        const at::Tensor& c_prev, const at::Tensor& c_out,
                        const at::Tensor& i_s, const at::Tensor& g_t,
                        const at::Tensor& f_s, const at::Tensor& o_s,
                        at::Tensor dgates_out, at::Tensor dc_prev_out);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rows_gather", &rows_gather, "embedding row gather (HIP)");
  m.def("segment_sum_rows", &segment_sum_rows,
        "indexed-slices segmented row sum (HIP)");
  m.def("multi_tensor_sgd", &multi_tensor_sgd);
  m.def("multi_tensor_momentum", &multi_tensor_momentum);
  m.def("multi_tensor_adagrad", &multi_tensor_adagrad);
  m.def("multi_tensor_adam", &multi_tensor_adam);
  m.def("multi_tensor_adam_dev", &multi_tensor_adam_dev);
  m.def("multi_tensor_rmsprop", &multi_tensor_rmsprop);
  m.def("multi_tensor_ema", &multi_tensor_ema);
  m.def("multi_tensor_zero", &multi_tensor_zero);
  m.def("px_attn_fwd", &px_attn_fwd);
  m.def("px_attn_bwd", &px_attn_bwd);
  m.def("multi_tensor_l2norm", &multi_tensor_l2norm);
  m.def("multi_tensor_scale_by_norm", &multi_tensor_scale_by_norm);
  m.def("sparse_adagrad", &sparse_adagrad, py::arg("param"),
        py::arg("accum"), py::arg("ids"), py::arg("grad_rows"),
        py::arg("lr"), py::arg("eps"), py::arg("nvalid") = py::none());
  m.def("sparse_sgd", &sparse_sgd, py::arg("param"), py::arg("ids"),
        py::arg("grad_rows"), py::arg("lr"),
        py::arg("nvalid") = py::none());
  m.def("multi_tensor_pack", &multi_tensor_pack);
  m.def("multi_tensor_unpack", &multi_tensor_unpack);
  m.def("batchnorm_fwd", &batchnorm_fwd);
  m.def("batchnorm_bwd", &batchnorm_bwd);
  m.def("gru_gates_fwd", &gru_gates_fwd);
  m.def("gru_gates_bwd", &gru_gates_bwd);
  m.def("maxpool3x3s2_fwd", &maxpool3x3s2_fwd);
  m.def("maxpool3x3s2_bwd", &maxpool3x3s2_bwd);
  m.def("gavgpool_fwd", &gavgpool_fwd);
  m.def("gavgpool_bwd", &gavgpool_bwd);
  m.def("col_sum", &col_sum,
        "fp32 column sum of a bf16 [M, N] matrix (bias-grad shape)");
  m.def("px_wgrad3x3", &px_wgrad3x3,
        "implicit-GEMM 3x3 NHWC conv wgrad (LDS-transposed staging, "
        "split-M fp32 atomic accumulation); returns fp32 [Cout, 9*Cin]");
  m.def("px_tall_conv3x3", &px_tall_conv3x3,
        "implicit-GEMM 3x3 NHWC conv fwd (and stride-1 dgrad via "
        "rotated weights) on the tall-tile structure");
  m.def("px_tall_bt", &px_tall_bt,
        "256x64-tile MFMA GEMM for M-huge/N-small (conv-as-GEMM) shapes");
  m.def("px_skinny_bt", &px_skinny_bt,
        "split-K skinny GEMM (M<=128) with fp32-atomic accumulation + "
        "fused cast/addend/re-zero epilogue (the LSTM chain launches)");
  m.def("pxgemm8_bt_v", &pxgemm8_bt_v,
        "pxgemm8_bt with explicit schedule-variant selection (A/B)");
  m.def("pxgemm8_bt", &pxgemm8_bt,
        "bf16 MFMA GEMM C=A@B^T, 256^2 8-phase deep-pipelined schedule");
  m.def("pxgemm8_bt_add", &pxgemm8_bt_add,
        "pxgemm8_bt with fused bf16 addend epilogue");
  m.def("pxgemm_bt", &pxgemm_bt,
        "hand-written MFMA bf16 GEMM C = A @ B^T");
  m.def("pxgemm_bt_add", &pxgemm_bt_add,
        "MFMA GEMM with fused bf16 addend epilogue");
  m.def("pxconv3x3_fwd", &pxconv3x3_fwd,
        "implicit-GEMM 3x3 NHWC conv forward (MFMA)");
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("softmax_ce_fwd", &softmax_ce_fwd);
  m.def("softmax_ce_bwd", &softmax_ce_bwd);
  m.def("sampled_ce_fwd", &sampled_ce_fwd);
  m.def("sampled_ce_bwd", &sampled_ce_bwd);
  m.def("lstm_gates_fwd", &lstm_gates_fwd);
  m.def("lstm_gates_bwd", &lstm_gates_bwd);
  m.def("lstm_gates_fwd_out", &lstm_gates_fwd_out);
  m.def("lstm_gates_bwd_out", &lstm_gates_bwd_out);
  m.def("lstm_step_fwd_fused", &lstm_step_fwd_fused,
        "fused LSTM step: recurrent gates GEMM + gate pointwise (MFMA)");
  m.def("lstm_step_bwd_fused", &lstm_step_bwd_fused,
        "fused LSTM step backward: ds GEMM + gate backward (MFMA)");
}
