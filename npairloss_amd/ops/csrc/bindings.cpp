// Python bindings for the npairloss_amd gfx950 HIP kernels.

#include <torch/extension.h>

#include <vector>

// npair_rows.hip
std::vector<torch::Tensor> rowstats(torch::Tensor S, torch::Tensor lab_l,
                                    torch::Tensor lab_g, int64_t rank);
std::vector<torch::Tensor> fused_fwd(torch::Tensor S, torch::Tensor lab_l,
                                     torch::Tensor lab_g, int64_t rank,
                                     torch::Tensor thr_p, torch::Tensor thr_n,
                                     torch::Tensor max_all, double margin_ident,
                                     double margin_diff, int64_t ap_method,
                                     int64_t an_method);
torch::Tensor bwd_weights(torch::Tensor S, torch::Tensor lab_l,
                          torch::Tensor lab_g, int64_t rank,
                          torch::Tensor thr_p, torch::Tensor thr_n,
                          torch::Tensor max_all, torch::Tensor loss_ident,
                          torch::Tensor loss_sum, double margin_ident,
                          double margin_diff, int64_t ap_method,
                          int64_t an_method, double scale);
torch::Tensor recall_hits(torch::Tensor S, torch::Tensor lab_l,
                          torch::Tensor lab_g, int64_t rank,
                          torch::Tensor ks_t, int64_t kmax);

// sort_select.hip
torch::Tensor local_relative_thr(torch::Tensor S, torch::Tensor lab_l,
                                 torch::Tensor lab_g, int64_t rank,
                                 bool use_same, double sn);
torch::Tensor global_relative_thr(torch::Tensor S, torch::Tensor lab_l,
                                  torch::Tensor lab_g, int64_t rank,
                                  bool use_same, double sn);

// l2norm.hip
std::vector<torch::Tensor> l2norm_fwd(torch::Tensor x);
torch::Tensor l2norm_bwd(torch::Tensor y, torch::Tensor inv_norm, torch::Tensor dy);

// vision.hip
torch::Tensor lrn_fwd(torch::Tensor x, int64_t size, double alpha, double beta, double k);
torch::Tensor lrn_bwd(torch::Tensor x, torch::Tensor dy, int64_t size, double alpha,
                      double beta, double k);
std::vector<torch::Tensor> maxpool3_fwd(torch::Tensor x, int64_t stride, bool ceil_mode);
torch::Tensor maxpool3_bwd(torch::Tensor dy, torch::Tensor idx, int64_t stride,
                           int64_t H, int64_t W);

// biasrelu.hip
torch::Tensor biasrelu_fwd(torch::Tensor x, torch::Tensor bias);
std::vector<torch::Tensor> biasrelu_bwd(torch::Tensor y, torch::Tensor dy);

// conv1x1.hip
torch::Tensor conv1x1_bias_relu_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias);
torch::Tensor conv1x1_dgrad(torch::Tensor g, torch::Tensor wt);

// gemm_lowp.hip
torch::Tensor sim_gemm_nt_bf16(torch::Tensor A, torch::Tensor B);
torch::Tensor sim_gemm_nt_fp8(torch::Tensor A, torch::Tensor B);
torch::Tensor cast_fp8(torch::Tensor x);

// gemm_f32.hip
torch::Tensor sim_gemm_nt(torch::Tensor F_l, torch::Tensor F_g);
torch::Tensor gemm_nn(torch::Tensor A, torch::Tensor B);
torch::Tensor gemm_tn(torch::Tensor A, torch::Tensor B);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "npairloss_amd gfx950 HIP kernels";
  m.def("rowstats", &rowstats, "per-query min_within/max_between/max_all");
  m.def("fused_fwd", &fused_fwd, "fused mask+select+count+LSE loss pass");
  m.def("bwd_weights", &bwd_weights, "fused (-p1+p2+p3) backward weights");
  m.def("recall_hits", &recall_hits, "Recall@k hit counts");
  m.def("local_relative_thr", &local_relative_thr, "per-row bitonic order-statistic threshold");
  m.def("global_relative_thr", &global_relative_thr, "device radix-select order-statistic threshold");
  m.def("l2norm_fwd", &l2norm_fwd, "row L2 normalize forward");
  m.def("l2norm_bwd", &l2norm_bwd, "row L2 normalize backward");
  m.def("lrn_fwd", &lrn_fwd, "fused across-channel LRN forward");
  m.def("lrn_bwd", &lrn_bwd, "fused across-channel LRN backward");
  m.def("maxpool3_fwd", &maxpool3_fwd, "3x3 max pool forward with argmax");
  m.def("maxpool3_bwd", &maxpool3_bwd, "3x3 max pool gather backward");
  m.def("sim_gemm_nt", &sim_gemm_nt, "fp32 MFMA similarity GEMM (A @ B^T)");
  m.def("sim_gemm_nt_bf16", &sim_gemm_nt_bf16, "bf16 MFMA similarity GEMM");
  m.def("sim_gemm_nt_fp8", &sim_gemm_nt_fp8, "fp8 e4m3 MFMA similarity GEMM");
  m.def("cast_fp8", &cast_fp8, "fp32 -> fp8 e4m3 bit pattern");
  m.def("biasrelu_fwd", &biasrelu_fwd, "fused bias+relu forward");
  m.def("biasrelu_bwd", &biasrelu_bwd, "fused drelu+bias-grad backward");
  m.def("conv1x1_bias_relu_fwd", &conv1x1_bias_relu_fwd,
        "fused 1x1-conv GEMM + bias + relu (bf16 MFMA)");
  m.def("conv1x1_dgrad", &conv1x1_dgrad, "1x1-conv data gradient GEMM (bf16 MFMA)");
  m.def("gemm_nn", &gemm_nn, "fp32 MFMA GEMM A @ B");
  m.def("gemm_tn", &gemm_tn, "fp32 MFMA GEMM A^T @ B");
}
