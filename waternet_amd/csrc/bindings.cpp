// Python bindings for the waternet_amd CDNA4 kernel library.
#include <torch/extension.h>

#include <vector>

// conv_mfma.hip
at::Tensor conv2d_fwd(const at::Tensor& x, const at::Tensor& wp,
                      const c10::optional<at::Tensor>& bias, int64_t ks,
                      int64_t Kp, int64_t Klog, int64_t act);
void conv2d_wgrad(const at::Tensor& dy, const at::Tensor& x, at::Tensor& dw,
                  int64_t ks);
void bias_grad(const at::Tensor& dy, at::Tensor& db);
at::Tensor pack_weight_fwd(const at::Tensor& w, int64_t Kp, int64_t Cp);
at::Tensor pack_weight_dgrad(const at::Tensor& w, int64_t Kp, int64_t Cp);
void pack_all(const at::Tensor& desc, int64_t njobs);
at::Tensor probe_tr();
at::Tensor probe_tr_raw();

// elementwise.hip
std::vector<at::Tensor> build_inputs(const at::Tensor& raw,
                                     const at::Tensor& wb,
                                     const at::Tensor& ce,
                                     const at::Tensor& gc);
at::Tensor nchw_to_nhwc(const at::Tensor& x, int64_t Cp);
at::Tensor nhwc_to_nchw(const at::Tensor& x, int64_t C);
at::Tensor fusion_fwd(const at::Tensor& maps, const at::Tensor& rwb,
                      const at::Tensor& rce, const at::Tensor& rgc);
std::vector<at::Tensor> fusion_bwd(const at::Tensor& dout,
                                   const at::Tensor& maps,
                                   const at::Tensor& rwb,
                                   const at::Tensor& rce,
                                   const at::Tensor& rgc);
at::Tensor act_bwd(const at::Tensor& dy, const at::Tensor& y, int64_t act);
at::Tensor act_bwd_bias(const at::Tensor& dy, const at::Tensor& y,
                        int64_t act, at::Tensor& db);
at::Tensor normalize_vgg_fwd(const at::Tensor& x, int64_t Cp);
at::Tensor normalize_vgg_bwd(const at::Tensor& dy, int64_t C);
at::Tensor sqdiff255_sum(const at::Tensor& a, const at::Tensor& b,
                         int64_t Clog);
at::Tensor sqdiff255_bwd(const at::Tensor& a, const at::Tensor& b,
                         const at::Tensor& gscale, double sign);
at::Tensor out_to_u8(const at::Tensor& x);
at::Tensor u8_to_nchw(const at::Tensor& x);
std::vector<at::Tensor> build_inputs_u8(const at::Tensor& raw,
                                        const at::Tensor& wb,
                                        const at::Tensor& ce,
                                        const at::Tensor& gc);
at::Tensor u8_to_nhwc(const at::Tensor& x, int64_t Cp);
at::Tensor normalize_nhwc_fwd(const at::Tensor& x);
at::Tensor normalize_nhwc_bwd(const at::Tensor& dy);
void adam_step(at::Tensor& p, const at::Tensor& g, at::Tensor& m,
               at::Tensor& v, const at::Tensor& lr_buf, double b1, double b2,
               double eps, at::Tensor& step_buf);

// pool.hip
std::vector<at::Tensor> maxpool2x2_fwd(const at::Tensor& x);
at::Tensor maxpool2x2_bwd(const at::Tensor& dy, const at::Tensor& idx,
                          int64_t H, int64_t W);

// ssim.hip
at::Tensor ssim_sum(const at::Tensor& a, const at::Tensor& b,
                    double data_range, double k1, double k2);
at::Tensor ssim_sum_nhwc(const at::Tensor& a, const at::Tensor& b,
                         int64_t Clog, double data_range, double k1,
                         double k2);

// preprocess.hip
std::vector<at::Tensor> preprocess_all(const at::Tensor& raw_u8);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("conv2d_fwd", &conv2d_fwd,
        "NHWC bf16 implicit-GEMM conv fwd (MFMA), fused bias+act");
  m.def("conv2d_wgrad", &conv2d_wgrad, "conv weight grad into NCHW fp32");
  m.def("bias_grad", &bias_grad, "bias grad (column sum)");
  m.def("pack_weight_fwd", &pack_weight_fwd);
  m.def("pack_weight_dgrad", &pack_weight_dgrad);
  m.def("pack_all", &pack_all, "batched fwd+dgrad weight repack");
  m.def("probe_tr", &probe_tr, "ds_read_b64_tr_b16 lane-mapping probe");
  m.def("probe_tr_raw", &probe_tr_raw, "raw tr permutation dump");
  m.def("build_inputs", &build_inputs);
  m.def("nchw_to_nhwc", &nchw_to_nhwc);
  m.def("nhwc_to_nchw", &nhwc_to_nchw);
  m.def("fusion_fwd", &fusion_fwd);
  m.def("fusion_bwd", &fusion_bwd);
  m.def("act_bwd", &act_bwd);
  m.def("act_bwd_bias", &act_bwd_bias,
        "activation backward fused with bias-grad column sums");
  m.def("normalize_vgg_fwd", &normalize_vgg_fwd);
  m.def("normalize_vgg_bwd", &normalize_vgg_bwd);
  m.def("sqdiff255_sum", &sqdiff255_sum);
  m.def("sqdiff255_bwd", &sqdiff255_bwd);
  m.def("adam_step", &adam_step);
  m.def("out_to_u8", &out_to_u8);
  m.def("u8_to_nchw", &u8_to_nchw);
  m.def("build_inputs_u8", &build_inputs_u8,
        "uint8 HWC batch -> cat-folded NHWC bf16 conv inputs");
  m.def("u8_to_nhwc", &u8_to_nhwc);
  m.def("normalize_nhwc_fwd", &normalize_nhwc_fwd);
  m.def("normalize_nhwc_bwd", &normalize_nhwc_bwd);
  m.def("maxpool2x2_fwd", &maxpool2x2_fwd);
  m.def("maxpool2x2_bwd", &maxpool2x2_bwd);
  m.def("ssim_sum", &ssim_sum);
  m.def("ssim_sum_nhwc", &ssim_sum_nhwc);
  m.def("preprocess_all", &preprocess_all);
}
