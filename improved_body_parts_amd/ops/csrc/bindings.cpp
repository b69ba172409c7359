// Python bindings for the CDNA4 kernel library.
#include <torch/extension.h>

using torch::Tensor;

// bn_act.hip
std::vector<Tensor> bn_stats(const Tensor& x_mc, int64_t C);
std::vector<Tensor> bn_finalize(
    const Tensor& sums, const Tensor& sumsq, int64_t M, const Tensor& gamma,
    const Tensor& beta, const c10::optional<Tensor>& running_mean,
    const c10::optional<Tensor>& running_var,
    const c10::optional<Tensor>& num_batches, double momentum, double eps);
std::vector<Tensor> bn_stats_finalize(
    const Tensor& x_mc, int64_t C, const Tensor& gamma, const Tensor& beta,
    const c10::optional<Tensor>& running_mean,
    const c10::optional<Tensor>& running_var,
    const c10::optional<Tensor>& num_batches, double momentum, double eps);
Tensor bn_act_fwd(const Tensor& x, const Tensor& scale, const Tensor& shift,
                  const c10::optional<Tensor>& residual, double slope, bool act);
std::vector<Tensor> bn_act_bwd(const Tensor& dy, const Tensor& y, const Tensor& x,
                               const c10::optional<Tensor>& mean,
                               const c10::optional<Tensor>& invstd,
                               double slope, bool act, bool need_xhat, int64_t C);
Tensor bn_act_bwd_apply(const Tensor& dpre, const Tensor& x, const Tensor& mean,
                        const Tensor& invstd, const Tensor& gamma,
                        const c10::optional<Tensor>& sum_dpre,
                        const c10::optional<Tensor>& sum_dxhat, int64_t C);

// spatial.hip
std::vector<Tensor> maxpool2x2_fwd(const Tensor& x, int64_t N, int64_t H,
                                   int64_t W, int64_t C);
Tensor maxpool2x2_bwd(const Tensor& dy, const Tensor& arg, int64_t N, int64_t H,
                      int64_t W, int64_t C);
Tensor upsample2x_fwd(const Tensor& x, int64_t N, int64_t H, int64_t W, int64_t C);
Tensor upsample2x_bwd(const Tensor& dy, int64_t N, int64_t H, int64_t W, int64_t C);
Tensor se_reduce(const Tensor& a, const c10::optional<Tensor>& b, int64_t N,
                 int64_t HW, int64_t C);
Tensor se_scale(const Tensor& x, const Tensor& s, const c10::optional<Tensor>& addc,
                int64_t N, int64_t HW, int64_t C);
Tensor se_gate(const Tensor& pooled, const Tensor& w1, const Tensor& b1,
               const Tensor& w2, const Tensor& b2, double slope,
               double pool_scale);

// loss.hip
Tensor focal_l2_fwd(const Tensor& pred, const Tensor& gt, const Tensor& mask,
                    int64_t heat_start, int64_t bkg_start, int64_t gamma,
                    double mtw, double ktw, double alpha, double beta);
Tensor focal_l2_bwd(const Tensor& pred, const Tensor& gt, const Tensor& mask,
                    const Tensor& stack_gscale, int64_t heat_start,
                    int64_t bkg_start, int64_t gamma, double mtw, double ktw,
                    double alpha, double beta);

// conv_mfma.hip
Tensor conv_mfma_fwd(const Tensor& x, const Tensor& w_packed, int64_t N,
                     int64_t H, int64_t W, int64_t Cin, int64_t Cout,
                     int64_t KH, int64_t KW, int64_t stride, int64_t pad_h,
                     int64_t pad_w, int64_t dil_h, int64_t dil_w, int64_t Ho,
                     int64_t Wo, const c10::optional<Tensor>& scale,
                     const c10::optional<Tensor>& shift,
                     const c10::optional<Tensor>& residual, bool act,
                     const c10::optional<Tensor>& residual_post,
                     const c10::optional<Tensor>& residual_post2, int64_t zs);

void pack_conv_weight(const Tensor& w, Tensor& fwd_pack,
                      const c10::optional<Tensor>& dgrad_pack);

// conv_wgrad.hip
Tensor conv_mfma_wgrad(const Tensor& x, const Tensor& dy, int64_t N, int64_t H,
                       int64_t W, int64_t Cin, int64_t Cout, int64_t KH,
                       int64_t KW, int64_t stride, int64_t pad_h, int64_t pad_w,
                       int64_t dil_h, int64_t dil_w, int64_t Ho, int64_t Wo);
Tensor tr16_probe();

// sgd.hip
void fused_sgd(const Tensor& chunk_table, double lr, double momentum,
               double weight_decay, int64_t dtype_tag);

// heatmap_gt.hip
Tensor heatmap_gt(const Tensor& joints, const c10::optional<Tensor>& mask_all,
                  const Tensor& limb_pairs, int64_t h, int64_t w,
                  int64_t stride, int64_t heat_start, int64_t bkg_start,
                  int64_t num_layers, double sigma, double paf_sigma,
                  double keypoint_thre, double limb_thre, double paf_thre);

// postproc.hip
Tensor heatmap_nms(const Tensor& heat, double thre);
std::vector<Tensor> collect_peaks(const Tensor& nmsed, const Tensor& smoothed,
                                  int64_t radius, int64_t max_peaks);
Tensor limb_scores(const Tensor& paf, const Tensor& peaks, const Tensor& cand_idx,
                   int64_t mid_num, double thre2);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("bn_stats", &bn_stats, "per-channel sum/sumsq of an [M][C] view");
  m.def("bn_finalize", &bn_finalize,
        "mean/invstd/scale/shift epilogue over given (possibly all-reduced) sums");
  m.def("bn_stats_finalize", &bn_stats_finalize,
        "stats + mean/invstd/scale/shift epilogue (+ running-stat update)");
  m.def("bn_act_fwd", &bn_act_fwd, "fused scale/shift (+res) (+leaky)");
  m.def("bn_act_bwd", &bn_act_bwd, "dpre + per-channel reductions");
  m.def("bn_act_bwd_apply", &bn_act_bwd_apply, "BN backward input grad");
  m.def("maxpool2x2_fwd", &maxpool2x2_fwd);
  m.def("maxpool2x2_bwd", &maxpool2x2_bwd);
  m.def("upsample2x_fwd", &upsample2x_fwd);
  m.def("upsample2x_bwd", &upsample2x_bwd);
  m.def("se_reduce", &se_reduce);
  m.def("se_scale", &se_scale);
  m.def("se_gate", &se_gate, "fused GAP-gate: sigmoid(W2@leaky(W1@p+b1)+b2)",
        py::arg("pooled"), py::arg("w1"), py::arg("b1"), py::arg("w2"),
        py::arg("b2"), py::arg("slope"), py::arg("pool_scale") = 1.0);
  m.def("focal_l2_fwd", &focal_l2_fwd);
  m.def("focal_l2_bwd", &focal_l2_bwd);
  m.def("conv_mfma_fwd", &conv_mfma_fwd, "MFMA implicit-GEMM conv forward",
        py::arg("x"), py::arg("w_packed"), py::arg("N"), py::arg("H"),
        py::arg("W"), py::arg("Cin"), py::arg("Cout"), py::arg("KH"),
        py::arg("KW"), py::arg("stride"), py::arg("pad_h"), py::arg("pad_w"),
        py::arg("dil_h"), py::arg("dil_w"), py::arg("Ho"), py::arg("Wo"),
        py::arg("scale") = c10::nullopt, py::arg("shift") = c10::nullopt,
        py::arg("residual") = c10::nullopt, py::arg("act") = false,
        py::arg("residual_post") = c10::nullopt,
        py::arg("residual_post2") = c10::nullopt, py::arg("zs") = 1);
  m.def("pack_conv_weight", &pack_conv_weight,
        "fwd + dgrad weight packs in one kernel",
        py::arg("w"), py::arg("fwd_pack"), py::arg("dgrad_pack") = c10::nullopt);
  m.def("conv_mfma_wgrad", &conv_mfma_wgrad,
        "MFMA implicit-GEMM conv weight gradient (tr16 LDS transpose)");
  m.def("tr16_probe", &tr16_probe,
        "debug: ds_read_b64_tr_b16 delivery-map probe");
  m.def("fused_sgd", &fused_sgd);
  m.def("heatmap_gt", &heatmap_gt, "on-device GT heatmap/PAF generation",
        py::arg("joints"), py::arg("mask_all"), py::arg("limb_pairs"),
        py::arg("h"), py::arg("w"), py::arg("stride"), py::arg("heat_start"),
        py::arg("bkg_start"), py::arg("num_layers"), py::arg("sigma"),
        py::arg("paf_sigma"), py::arg("keypoint_thre"), py::arg("limb_thre"),
        py::arg("paf_thre"));
  m.def("heatmap_nms", &heatmap_nms);
  m.def("collect_peaks", &collect_peaks);
  m.def("limb_scores", &limb_scores);
}
