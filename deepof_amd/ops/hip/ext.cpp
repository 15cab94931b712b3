// Python bindings for the deepof_amd gfx950 HIP extension.

#include <torch/extension.h>

at::Tensor warp_forward(at::Tensor img2, at::Tensor flow);
std::vector<at::Tensor> warp_backward(at::Tensor grad_out, at::Tensor img2,
                                      at::Tensor flow);
std::vector<at::Tensor> unsup_loss_forward(at::Tensor flow, at::Tensor img1,
                                           at::Tensor img2, double scale,
                                           double eps, double alpha_c,
                                           double alpha_s, bool want_recon);
at::Tensor unsup_loss_backward(at::Tensor flow, at::Tensor img1,
                               at::Tensor img2, double scale, double eps,
                               double alpha_c, double alpha_s, at::Tensor gs);
at::Tensor resize_bilinear(at::Tensor x, long oh, long ow);
at::Tensor lrn_forward(at::Tensor x, long radius, double bias, double alpha,
                       double beta);
at::Tensor epe_sum(at::Tensor f, at::Tensor g);
at::Tensor act_grad(at::Tensor gy, at::Tensor y, long act);
at::Tensor correlation_forward(at::Tensor f1, at::Tensor f2, long md);
std::vector<at::Tensor> correlation_backward(at::Tensor gout, at::Tensor f1,
                                             at::Tensor f2, long md);
at::Tensor conv2d_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                      long stride, long pad, long act);
at::Tensor conv2d_fwd256(at::Tensor x, at::Tensor w, at::Tensor bias,
                         long stride, long pad, long act);
void conv2d_fwd_strided(at::Tensor x, at::Tensor w, at::Tensor bias,
                        at::Tensor out, long pad_y, long pad_x, long act,
                        long ostride, long off_y, long off_x,
                        long out_coff);
at::Tensor subpixel_pack(at::Tensor w, at::Tensor tab, long R, long S);
void conv2d_fwd_subpixel4(at::Tensor x, at::Tensor wpacked, at::Tensor bias,
                          at::Tensor out, at::Tensor ptab, long K, long act,
                          long ostride, long out_coff);
at::Tensor conv2d_wrw2(at::Tensor gy, at::Tensor x, long R, long S,
                       long stride, long pad);
at::Tensor build_adam_table(std::vector<at::Tensor> params,
                            std::vector<at::Tensor> grads,
                            std::vector<at::Tensor> exp_avgs,
                            std::vector<at::Tensor> exp_avg_sqs);
void fused_adam_table(at::Tensor table, long n_chunks, at::Tensor hyper,
                      double beta1, double beta2, double eps, double wd);
void fused_adam(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                std::vector<at::Tensor> exp_avgs,
                std::vector<at::Tensor> exp_avg_sqs, double lr, double beta1,
                double beta2, double eps, double wd, double bias1,
                double bias2);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("warp_forward", &warp_forward, "bilinear backward warp (fwd)");
  m.def("warp_backward", &warp_backward, "bilinear backward warp (bwd)");
  m.def("unsup_loss_forward", &unsup_loss_forward,
        "fused warp+Charbonnier+smoothness (fwd sums)");
  m.def("unsup_loss_backward", &unsup_loss_backward,
        "fused loss backward -> d(flow)");
  m.def("resize_bilinear", &resize_bilinear, "legacy-TF bilinear resize");
  m.def("lrn_forward", &lrn_forward, "across-channel LRN");
  m.def("epe_sum", &epe_sum, "endpoint-error sum reduction");
  m.def("act_grad", &act_grad, "fused activation gradient from output");
  m.def("correlation_forward", &correlation_forward, "cost volume fwd");
  m.def("correlation_backward", &correlation_backward, "cost volume bwd");
  m.def("fused_adam", &fused_adam, "multi-tensor Adam step");
  m.def("build_adam_table", &build_adam_table, "pack Adam chunk table");
  m.def("fused_adam_table", &fused_adam_table, "Adam step from table");
  m.def("conv2d_fwd", &conv2d_fwd,
        "MFMA implicit-GEMM conv + bias + act (NHWC bf16)");
  m.def("conv2d_fwd256", &conv2d_fwd256,
        "deep-pipelined 256x256 MFMA conv (counted vmcnt, raw barriers)");
  m.def("conv2d_fwd_strided", &conv2d_fwd_strided,
        "sub-pixel strided-output conv (deconv fwd / stride-2 bwd-data "
        "parity launch)");
  m.def("conv2d_wrw2", &conv2d_wrw2,
        "MFMA weight gradient v2 (natural-layout staging, tr_b16 reads)");
  m.def("subpixel_pack", &subpixel_pack,
        "gather all 4 parity sub-filters of a transposed conv");
  m.def("conv2d_fwd_subpixel4", &conv2d_fwd_subpixel4,
        "all-parity sub-pixel transposed conv in one launch");
}
