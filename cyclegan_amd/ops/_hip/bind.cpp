// pybind registration for the gfx950 kernel library.
#include <torch/extension.h>

namespace cyg {
at::Tensor conv2d_fwd(at::Tensor, at::Tensor, c10::optional<at::Tensor>,
                      int64_t, int64_t, int64_t, int64_t, int64_t, bool,
                      int64_t, double);
at::Tensor convt2d_fwd(at::Tensor, at::Tensor, c10::optional<at::Tensor>,
                       int64_t, int64_t, int64_t, int64_t, int64_t, int64_t,
                       double);
at::Tensor conv2d_dgrad(at::Tensor, at::Tensor, int64_t, int64_t, int64_t,
                        int64_t, int64_t, int64_t, int64_t, bool);
at::Tensor convt2d_dgrad(at::Tensor, at::Tensor, int64_t, int64_t, int64_t,
                         int64_t, int64_t);
at::Tensor conv2d_wgrad(at::Tensor, at::Tensor, int64_t, int64_t, int64_t,
                        int64_t, int64_t, bool);
at::Tensor mfma_probe(at::Tensor, at::Tensor);
at::Tensor mx_probe(at::Tensor, at::Tensor, int64_t, int64_t);
at::Tensor mx_probe16(at::Tensor, at::Tensor, int64_t, int64_t);
at::Tensor glds_probe(at::Tensor, at::Tensor);
at::Tensor tr_probe(int64_t);
at::Tensor quant_fp8(at::Tensor, at::Tensor);
at::Tensor quant_fp8_d(at::Tensor, at::Tensor, at::Tensor);
void amax_roll(at::Tensor, int64_t);
at::Tensor conv2d_fp8_fwd(at::Tensor, at::Tensor, at::Tensor,
                          c10::optional<at::Tensor>, c10::optional<at::Tensor>,
                          int64_t, int64_t, int64_t, int64_t, int64_t, bool, int64_t, double);
std::vector<at::Tensor> instnorm_fwd(at::Tensor, at::Tensor, at::Tensor,
                                     double, int64_t, double,
                                     c10::optional<at::Tensor>);
std::vector<at::Tensor> instnorm_bwd(at::Tensor, at::Tensor, at::Tensor,
                                     at::Tensor, at::Tensor,
                                     c10::optional<at::Tensor>, int64_t,
                                     double);
at::Tensor act_bwd(at::Tensor, at::Tensor, int64_t, double);
at::Tensor channel_sum(at::Tensor);
at::Tensor reflect_pad_fwd(at::Tensor, int64_t, int64_t, int64_t, int64_t);
at::Tensor reflect_pad_bwd(at::Tensor, int64_t, int64_t, int64_t, int64_t);
at::Tensor persample_loss_fwd(at::Tensor, at::Tensor, bool);
at::Tensor persample_loss_const_fwd(at::Tensor, double, bool);
std::vector<at::Tensor> persample_loss_bwd(at::Tensor, at::Tensor, at::Tensor,
                                           bool, bool, bool);
at::Tensor persample_loss_const_bwd(at::Tensor, double, at::Tensor, bool);
void adam_step_dev(at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor,
                   double, double, double);
void shadow_gather(at::Tensor, at::Tensor, at::Tensor);
void adam_step(at::Tensor, at::Tensor, at::Tensor, at::Tensor, double, double,
               double, double, int64_t);
}  // namespace cyg

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("conv2d_fwd", &cyg::conv2d_fwd);
  m.def("convt2d_fwd", &cyg::convt2d_fwd);
  m.def("conv2d_dgrad", &cyg::conv2d_dgrad);
  m.def("convt2d_dgrad", &cyg::convt2d_dgrad);
  m.def("conv2d_wgrad", &cyg::conv2d_wgrad);
  m.def("mfma_probe", &cyg::mfma_probe);
  m.def("mx_probe", &cyg::mx_probe);
  m.def("mx_probe16", &cyg::mx_probe16);
  m.def("glds_probe", &cyg::glds_probe);
  m.def("tr_probe", &cyg::tr_probe);
  m.def("quant_fp8", &cyg::quant_fp8);
  m.def("quant_fp8_d", &cyg::quant_fp8_d);
  m.def("amax_roll", &cyg::amax_roll);
  m.def("conv2d_fp8_fwd", &cyg::conv2d_fp8_fwd);
  m.def("instnorm_fwd", &cyg::instnorm_fwd);
  m.def("instnorm_bwd", &cyg::instnorm_bwd);
  m.def("act_bwd", &cyg::act_bwd);
  m.def("channel_sum", &cyg::channel_sum);
  m.def("reflect_pad_fwd", &cyg::reflect_pad_fwd);
  m.def("reflect_pad_bwd", &cyg::reflect_pad_bwd);
  m.def("persample_loss_fwd", &cyg::persample_loss_fwd);
  m.def("persample_loss_const_fwd", &cyg::persample_loss_const_fwd);
  m.def("persample_loss_bwd", &cyg::persample_loss_bwd);
  m.def("persample_loss_const_bwd", &cyg::persample_loss_const_bwd);
  m.def("adam_step", &cyg::adam_step);
  m.def("adam_step_dev", &cyg::adam_step_dev);
  m.def("shadow_gather", &cyg::shadow_gather);
}
