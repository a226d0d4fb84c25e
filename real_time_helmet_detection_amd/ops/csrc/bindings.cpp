// pybind bindings for the gfx950 kernel set (real_time_helmet_detection_amd.ops._C)
#include <torch/extension.h>

namespace rthd {
torch::Tensor add_act_fwd(torch::Tensor a, torch::Tensor b, int64_t act);
torch::Tensor add_act_bwd(torch::Tensor dy, torch::Tensor y, int64_t act);

std::vector<torch::Tensor> pool2x2_fwd(torch::Tensor x, bool is_max,
                                       bool need_arg);
torch::Tensor pool2x2_bwd(torch::Tensor dy, torch::Tensor arg, bool is_max,
                          int64_t H, int64_t W);
std::vector<torch::Tensor> maxpool_same_fwd(torch::Tensor x, int64_t k,
                                            bool need_arg);
torch::Tensor maxpool_same_bwd(torch::Tensor dy, torch::Tensor arg,
                               int64_t k);

torch::Tensor upsample2x_add_fwd(torch::Tensor x,
                                 c10::optional<torch::Tensor> skip);
torch::Tensor upsample2x_bwd(torch::Tensor dy);

std::vector<torch::Tensor> centernet_loss_fwd(
    torch::Tensor phm, torch::Tensor poff, torch::Tensor psize,
    torch::Tensor ghm, torch::Tensor goff, torch::Tensor gsize,
    torch::Tensor mask, double alpha, double beta);
std::vector<torch::Tensor> centernet_loss_bwd(
    torch::Tensor phm, torch::Tensor poff, torch::Tensor psize,
    torch::Tensor ghm, torch::Tensor goff, torch::Tensor gsize,
    torch::Tensor mask, torch::Tensor sums, torch::Tensor gout,
    double alpha, double beta);
std::vector<torch::Tensor> centernet_loss_fused_fwd(
    torch::Tensor out, torch::Tensor ghm, torch::Tensor goff,
    torch::Tensor gsize, torch::Tensor mask, double alpha, double beta,
    bool sig_os);
torch::Tensor centernet_loss_fused_bwd(
    torch::Tensor out, torch::Tensor ghm, torch::Tensor goff,
    torch::Tensor gsize, torch::Tensor mask, torch::Tensor sums,
    torch::Tensor glosses, double alpha, double beta, bool sig_os);

std::vector<torch::Tensor> decode_fwd(torch::Tensor hm, torch::Tensor off,
                                      torch::Tensor wh, int64_t scale_factor,
                                      int64_t topk, int64_t pool_size,
                                      bool normalized);
torch::Tensor nms_fwd(torch::Tensor boxes, torch::Tensor scores,
                      double iou_threshold);
std::vector<torch::Tensor> nms_batched(torch::Tensor boxes,
                                       torch::Tensor scores,
                                       double iou_threshold,
                                       double conf_th);

torch::Tensor pack_weights(torch::Tensor w, bool swap, bool to_bf16);
torch::Tensor pack_weights_fp8(torch::Tensor w);
torch::Tensor conv_fwd_fp8r(torch::Tensor x, torch::Tensor wpk,
                            torch::Tensor scale, torch::Tensor shift,
                            c10::optional<torch::Tensor> skip,
                            int64_t KH, int64_t KW, int64_t stride,
                            int64_t pad, int64_t Cout, int64_t act,
                            bool out_fp8);
torch::Tensor conv_fwd(torch::Tensor x, torch::Tensor wpk,
                       torch::Tensor scale, torch::Tensor shift,
                       c10::optional<torch::Tensor> skip,
                       int64_t KH, int64_t KW, int64_t stride, int64_t pad,
                       int64_t Cout, int64_t act);
torch::Tensor conv_fwd_small(torch::Tensor x, torch::Tensor wpk,
                             torch::Tensor scale, torch::Tensor shift,
                             c10::optional<torch::Tensor> skip,
                             int64_t KH, int64_t KW, int64_t stride,
                             int64_t pad, int64_t Cout, int64_t act,
                             int64_t splitk);
torch::Tensor conv_fwd_k64(torch::Tensor x, torch::Tensor wpk,
                           torch::Tensor scale, torch::Tensor shift,
                           c10::optional<torch::Tensor> skip,
                           int64_t KH, int64_t KW, int64_t stride,
                           int64_t pad, int64_t Cout, int64_t act);
torch::Tensor stem_im2col(torch::Tensor x, int64_t KS, int64_t stride,
                          int64_t pad);
torch::Tensor wgrad(torch::Tensor x, torch::Tensor dy, int64_t KH,
                    int64_t KW, int64_t stride, int64_t pad);
torch::Tensor wgrad_bf16_fast(torch::Tensor x, torch::Tensor dy, int64_t KH,
                              int64_t KW, int64_t stride, int64_t pad);

std::vector<torch::Tensor> conv_fwd_stats(
    torch::Tensor x, torch::Tensor wpk, torch::Tensor scale,
    torch::Tensor shift, int64_t KH, int64_t KW, int64_t stride,
    int64_t pad, int64_t Cout, int64_t act);
std::vector<torch::Tensor> bn_stats_from_parts(
    torch::Tensor p1, torch::Tensor p2,
    c10::optional<torch::Tensor> running_mean,
    c10::optional<torch::Tensor> running_var,
    double momentum, double eps, int64_t M);
std::vector<torch::Tensor> bn_stats(torch::Tensor x,
                                    c10::optional<torch::Tensor> running_mean,
                                    c10::optional<torch::Tensor> running_var,
                                    double momentum, double eps);
torch::Tensor bn_act_fwd(torch::Tensor x, torch::Tensor mean,
                         torch::Tensor rstd, torch::Tensor gamma,
                         torch::Tensor beta, int64_t act,
                         c10::optional<torch::Tensor> skip);
std::vector<torch::Tensor> bn_act_bwd(torch::Tensor dy, torch::Tensor x,
                                      torch::Tensor mean, torch::Tensor rstd,
                                      torch::Tensor gamma, torch::Tensor beta,
                                      int64_t act,
                                      c10::optional<torch::Tensor> skip);
torch::Tensor col_sum(torch::Tensor x);
}  // namespace rthd

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("add_act_fwd", &rthd::add_act_fwd);
  m.def("add_act_bwd", &rthd::add_act_bwd);
  m.def("pool2x2_fwd", &rthd::pool2x2_fwd);
  m.def("pool2x2_bwd", &rthd::pool2x2_bwd);
  m.def("maxpool_same_fwd", &rthd::maxpool_same_fwd);
  m.def("maxpool_same_bwd", &rthd::maxpool_same_bwd);
  m.def("upsample2x_add_fwd", &rthd::upsample2x_add_fwd);
  m.def("upsample2x_bwd", &rthd::upsample2x_bwd);
  m.def("centernet_loss_fwd", &rthd::centernet_loss_fwd);
  m.def("centernet_loss_bwd", &rthd::centernet_loss_bwd);
  m.def("centernet_loss_fused_fwd", &rthd::centernet_loss_fused_fwd);
  m.def("centernet_loss_fused_bwd", &rthd::centernet_loss_fused_bwd);
  m.def("decode_fwd", &rthd::decode_fwd);
  m.def("nms_fwd", &rthd::nms_fwd);
  m.def("nms_batched", &rthd::nms_batched);
  m.def("pack_weights", &rthd::pack_weights);
  m.def("pack_weights_fp8", &rthd::pack_weights_fp8);
  m.def("conv_fwd_fp8r", &rthd::conv_fwd_fp8r);
  m.def("conv_fwd", &rthd::conv_fwd);
  m.def("conv_fwd_small", &rthd::conv_fwd_small);
  m.def("conv_fwd_k64", &rthd::conv_fwd_k64);
  m.def("stem_im2col", &rthd::stem_im2col);
  m.def("wgrad", &rthd::wgrad);
  m.def("wgrad_bf16_fast", &rthd::wgrad_bf16_fast);
  m.def("conv_fwd_stats", &rthd::conv_fwd_stats);
  m.def("bn_stats_from_parts", &rthd::bn_stats_from_parts,
        py::arg("p1"), py::arg("p2"), py::arg("running_mean") = py::none(),
        py::arg("running_var") = py::none(), py::arg("momentum") = 0.1,
        py::arg("eps") = 1e-5, py::arg("m") = 1);
  m.def("bn_stats", &rthd::bn_stats);
  m.def("bn_act_fwd", &rthd::bn_act_fwd, py::arg("x"), py::arg("mean"), py::arg("rstd"), py::arg("gamma"), py::arg("beta"), py::arg("act"), py::arg("skip") = py::none());
  m.def("bn_act_bwd", &rthd::bn_act_bwd, py::arg("dy"), py::arg("x"), py::arg("mean"), py::arg("rstd"), py::arg("gamma"), py::arg("beta"), py::arg("act"), py::arg("skip") = py::none());
  m.def("col_sum", &rthd::col_sum);
}

// ---------------------------------------------------------------------------
// Dispatcher registration (torch.ops.rthd.*) for the inference-path ops.
//
// torch.jit.trace can only record dispatcher ops, so the GPU export path
// (engine/exporter.py) calls these instead of the pybind entry points: the
// traced jit_traced_model_gpu.pth then embeds the native gfx950 kernels and
// the C++ app (tools/cpp_infer, -k flag) runs them via LibTorch with no
// python. The pools get tensor-only wrappers (the pybind variants return
// [y, argmax] for autograd; inference needs just y).
// ---------------------------------------------------------------------------
namespace rthd {

static torch::Tensor maxpool2x2_op(torch::Tensor x) {
  return pool2x2_fwd(std::move(x), /*is_max=*/true, /*need_arg=*/false)[0];
}
static torch::Tensor avgpool2x2_op(torch::Tensor x) {
  return pool2x2_fwd(std::move(x), /*is_max=*/false, /*need_arg=*/false)[0];
}
static torch::Tensor maxpool_same_op(torch::Tensor x, int64_t k) {
  return maxpool_same_fwd(std::move(x), k, /*need_arg=*/false)[0];
}
static std::tuple<torch::Tensor, torch::Tensor> nms_batched_op(
    torch::Tensor boxes, torch::Tensor scores, double iou, double conf) {
  auto v = nms_batched(std::move(boxes), std::move(scores), iou, conf);
  return {v[0], v[1]};
}
static std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> decode_op(
    torch::Tensor hm, torch::Tensor off, torch::Tensor wh,
    int64_t scale_factor, int64_t topk, int64_t pool_size, bool normalized) {
  auto v = decode_fwd(std::move(hm), std::move(off), std::move(wh),
                      scale_factor, topk, pool_size, normalized);
  return {v[0], v[1], v[2]};
}

}  // namespace rthd

TORCH_LIBRARY(rthd, m) {
  m.def("conv_fwd(Tensor x, Tensor wpk, Tensor scale, Tensor shift, "
        "Tensor? skip, int kh, int kw, int stride, int pad, int cout, "
        "int act) -> Tensor");
  m.def("conv_fwd_fp8r(Tensor x, Tensor wpk, Tensor scale, Tensor shift, "
        "Tensor? skip, int kh, int kw, int stride, int pad, int cout, "
        "int act, bool out_fp8) -> Tensor");
  m.def("add_act_fwd(Tensor a, Tensor b, int act) -> Tensor");
  m.def("maxpool2x2(Tensor x) -> Tensor");
  m.def("avgpool2x2(Tensor x) -> Tensor");
  m.def("maxpool_same(Tensor x, int k) -> Tensor");
  m.def("upsample2x_add(Tensor x, Tensor? skip) -> Tensor");
  m.def("nms(Tensor boxes, Tensor scores, float iou) -> Tensor");
  m.def("nms_batched(Tensor boxes, Tensor scores, float iou, float conf) "
        "-> (Tensor, Tensor)");
  m.def("decode(Tensor hm, Tensor off, Tensor wh, int scale_factor, "
        "int topk, int pool_size, bool normalized) "
        "-> (Tensor, Tensor, Tensor)");
  m.def("stem_im2col(Tensor x, int ks, int stride, int pad) -> Tensor");
}

TORCH_LIBRARY_IMPL(rthd, CUDA, m) {
  m.impl("conv_fwd", rthd::conv_fwd);
  m.impl("conv_fwd_fp8r", rthd::conv_fwd_fp8r);
  m.impl("add_act_fwd", rthd::add_act_fwd);
  m.impl("maxpool2x2", rthd::maxpool2x2_op);
  m.impl("avgpool2x2", rthd::avgpool2x2_op);
  m.impl("maxpool_same", rthd::maxpool_same_op);
  m.impl("upsample2x_add", rthd::upsample2x_add_fwd);
  m.impl("nms", rthd::nms_fwd);
  m.impl("nms_batched", rthd::nms_batched_op);
  m.impl("decode", rthd::decode_op);
  m.impl("stem_im2col", rthd::stem_im2col);
}
