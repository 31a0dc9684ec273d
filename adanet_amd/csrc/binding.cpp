// pybind11 bindings for the adanet_amd CDNA4 kernel library (_adanet_hip).
// Built in-tree with torch.utils.cpp_extension under PYTORCH_ROCM_ARCH=gfx950
// (see setup.py / __graft_entry__.build()).

#include <torch/extension.h>

void gemm_nt_bf16(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
                  const c10::optional<at::Tensor>& bias, int64_t act,
                  double dropout_p, const c10::optional<at::Tensor>& seed);
void gemm_nt_bf16_probe(const at::Tensor& A, const at::Tensor& B,
                        at::Tensor& C, int64_t variant);
void gemm_nt8(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
              const c10::optional<at::Tensor>& bias, int64_t act,
              int64_t variant);
void gemm_x8(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
             const c10::optional<at::Tensor>& bias, int64_t act,
             int64_t trans_a, int64_t trans_b, int64_t variant);
void probe_tr16_layout(at::Tensor& out);
void gemm_tr_probe(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
                   int64_t trans_a, int64_t trans_b, int64_t variant);
void gemm_tr_bf16(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
                  const c10::optional<at::Tensor>& bias, int64_t act,
                  int64_t trans_a, int64_t trans_b);
void gemm_tr_batched(const at::Tensor& A, const at::Tensor& B, at::Tensor& C,
                     const c10::optional<at::Tensor>& bias, int64_t act,
                     int64_t trans_a, int64_t trans_b);
void transpose_bf16(const at::Tensor& in, at::Tensor& out);
void softmax_xent_fwd(const at::Tensor& logits, const at::Tensor& labels,
                      const c10::optional<at::Tensor>& loss,
                      at::Tensor& probs, double eps,
                      const c10::optional<at::Tensor>& mean_out);
void softmax_xent_bwd(const at::Tensor& probs, const at::Tensor& labels,
                      const c10::optional<at::Tensor>& grad_rows,
                      at::Tensor& dlogits, double eps,
                      const c10::optional<at::Tensor>& grad_scalar);
void mixer_fwd(const at::Tensor& stack, const at::Tensor& weights,
               const c10::optional<at::Tensor>& bias, at::Tensor& out,
               int64_t vector_mode);
void mixer_bwd_dw(const at::Tensor& stack, const at::Tensor& dY,
                  at::Tensor& dw, int64_t vector_mode);
void mixer_bwd_dlogits(const at::Tensor& dY, const at::Tensor& weights,
                       at::Tensor& dL, int64_t j, int64_t vector_mode);
void mixer_fwd_direct(const std::vector<at::Tensor>& members,
                      const at::Tensor& weights,
                      const c10::optional<at::Tensor>& bias, at::Tensor& out,
                      int64_t vector_mode);
void mixer_bwd_dw_direct(const std::vector<at::Tensor>& members,
                         const at::Tensor& dY, at::Tensor& dw,
                         int64_t vector_mode);
void fused_sgd(at::Tensor& master, at::Tensor& param, const at::Tensor& grad,
               const c10::optional<at::Tensor>& momentum_buf, double lr,
               double momentum, double dampening, double weight_decay,
               bool nesterov, double grad_scale);
void fused_adam(at::Tensor& master, at::Tensor& param, const at::Tensor& grad,
                at::Tensor& m_buf, at::Tensor& v_buf, double lr, double beta1,
                double beta2, double eps, double weight_decay, int64_t step,
                double grad_scale);
void fused_sgd_fp32(at::Tensor& param, const at::Tensor& grad,
                    const c10::optional<at::Tensor>& momentum_buf, double lr,
                    double momentum, double dampening, double weight_decay,
                    bool nesterov, double grad_scale);
void fused_adam_fp32(at::Tensor& param, const at::Tensor& grad,
                     at::Tensor& m_buf, at::Tensor& v_buf, double lr,
                     double beta1, double beta2, double eps,
                     double weight_decay, int64_t step, double grad_scale);
void layernorm_fwd(const at::Tensor& x, const c10::optional<at::Tensor>& gamma,
                   const c10::optional<at::Tensor>& beta, at::Tensor& y,
                   at::Tensor& mean, at::Tensor& rstd, double eps);
void layernorm_bwd(const at::Tensor& x, const at::Tensor& dy,
                   const c10::optional<at::Tensor>& gamma,
                   const at::Tensor& mean, const at::Tensor& rstd,
                   at::Tensor& dx, const c10::optional<at::Tensor>& dgamma,
                   const c10::optional<at::Tensor>& dbeta);
void dropout_fwd(const at::Tensor& x, at::Tensor& y, double p,
                 const at::Tensor& seed);
void dropout_bwd(const at::Tensor& dy, at::Tensor& dx, double p,
                 const at::Tensor& seed);
void relu_bwd(const at::Tensor& dy, const at::Tensor& y, at::Tensor& dx,
              double scale);
void relu_bwd_colsum(const at::Tensor& dy, const at::Tensor& y,
                     at::Tensor& dz, at::Tensor& db, double scale);
void multi_copy_bf16(const std::vector<at::Tensor>& srcs,
                     const std::vector<at::Tensor>& dsts);
void colsum_bf16(const at::Tensor& x, at::Tensor& out, int64_t accum);
void binary_histogram(const at::Tensor& scores, const at::Tensor& labels,
                      at::Tensor& hist);
void depthwise_fwd(const at::Tensor& x, const at::Tensor& w, at::Tensor& y,
                   int64_t stride, int64_t pad);
void depthwise_bwd_dx(const at::Tensor& dy, const at::Tensor& w,
                      at::Tensor& dx, int64_t stride, int64_t pad);
void depthwise_bwd_dw(const at::Tensor& x, const at::Tensor& dy,
                      at::Tensor& dw, int64_t stride, int64_t pad);
void argmax_correct(const at::Tensor& logits, const at::Tensor& labels,
                    const c10::optional<at::Tensor>& pred, at::Tensor& correct);
void im2col_bf16(const at::Tensor& x, at::Tensor& out, int64_t K,
                 int64_t stride, int64_t pad);
void col2im_bf16(const at::Tensor& du, at::Tensor& dx, int64_t K,
                 int64_t stride, int64_t pad);
void pool3_fwd(const at::Tensor& x, at::Tensor& y,
               const c10::optional<at::Tensor>& argmax, int64_t stride,
               int64_t is_max);
void pool3_bwd(const at::Tensor& dy, const c10::optional<at::Tensor>& argmax,
               at::Tensor& dx, int64_t stride, int64_t is_max);
void batchnorm_stats(const at::Tensor& x, at::Tensor& mean, at::Tensor& rstd,
                     const c10::optional<at::Tensor>& running_mean,
                     const c10::optional<at::Tensor>& running_var,
                     double eps, double momentum);
void batchnorm_norm(const at::Tensor& x, at::Tensor& y,
                    const at::Tensor& mean, const at::Tensor& rstd,
                    const c10::optional<at::Tensor>& gamma,
                    const c10::optional<at::Tensor>& beta);
void batchnorm_bwd(const at::Tensor& x, const at::Tensor& dy, at::Tensor& dx,
                   const at::Tensor& mean, const at::Tensor& rstd,
                   const c10::optional<at::Tensor>& gamma, at::Tensor& sdy,
                   at::Tensor& sdyx);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "adanet_amd gfx950 (MI355X/CDNA4) kernels";
  m.def("gemm_nt_bf16", &gemm_nt_bf16,
        "C[M,N] = A[M,K] @ B[N,K]^T, fused bias/relu/dropout",
        py::arg("A"), py::arg("B"), py::arg("C"), py::arg("bias"),
        py::arg("act"), py::arg("dropout_p") = 0.0,
        py::arg("seed") = c10::nullopt);
  m.def("gemm_nt_bf16_probe", &gemm_nt_bf16_probe);
  m.def("gemm_nt8", &gemm_nt8,
        "8-phase deep-pipelined GEMM (256^2-class tiles)");
  m.def("gemm_x8", &gemm_x8,
        "pipelined GEMM with K-major (transposed) operand staging");
  m.def("probe_tr16_layout", &probe_tr16_layout);
  m.def("gemm_tr_probe", &gemm_tr_probe);
  m.def("gemm_tr_bf16", &gemm_tr_bf16);
  m.def("gemm_tr_batched", &gemm_tr_batched);
  m.def("transpose_bf16", &transpose_bf16, "bf16 2-D transpose");
  m.def("softmax_xent_fwd", &softmax_xent_fwd);
  m.def("softmax_xent_bwd", &softmax_xent_bwd);
  m.def("mixer_fwd", &mixer_fwd);
  m.def("mixer_bwd_dw", &mixer_bwd_dw);
  m.def("mixer_bwd_dlogits", &mixer_bwd_dlogits);
  m.def("mixer_fwd_direct", &mixer_fwd_direct);
  m.def("mixer_bwd_dw_direct", &mixer_bwd_dw_direct);
  m.def("fused_sgd", &fused_sgd);
  m.def("fused_adam", &fused_adam);
  m.def("fused_sgd_fp32", &fused_sgd_fp32);
  m.def("fused_adam_fp32", &fused_adam_fp32);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("relu_bwd", &relu_bwd, py::arg("dy"), py::arg("y"), py::arg("dx"),
        py::arg("scale") = 1.0);
  m.def("relu_bwd_colsum", &relu_bwd_colsum, py::arg("dy"), py::arg("y"),
        py::arg("dz"), py::arg("db"), py::arg("scale") = 1.0);
  m.def("multi_copy_bf16", &multi_copy_bf16);
  m.def("colsum_bf16", &colsum_bf16, py::arg("x"), py::arg("out"),
        py::arg("accum") = 0);
  m.def("argmax_correct", &argmax_correct);
  m.def("binary_histogram", &binary_histogram);
  m.def("im2col_bf16", &im2col_bf16);
  m.def("col2im_bf16", &col2im_bf16);
  m.def("pool3_fwd", &pool3_fwd);
  m.def("pool3_bwd", &pool3_bwd);
  m.def("batchnorm_stats", &batchnorm_stats);
  m.def("batchnorm_norm", &batchnorm_norm);
  m.def("batchnorm_bwd", &batchnorm_bwd);
  m.def("depthwise_fwd", &depthwise_fwd);
  m.def("depthwise_bwd_dx", &depthwise_bwd_dx);
  m.def("depthwise_bwd_dw", &depthwise_bwd_dw);
}
