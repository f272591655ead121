// Python bindings for the fast_autoaugment_amd CDNA4 kernel library.
#include <torch/extension.h>

torch::Tensor scale_bcast(torch::Tensor x, torch::Tensor s);
torch::Tensor scale_lerp(torch::Tensor x1, torch::Tensor x2, torch::Tensor a);
torch::Tensor swish_fwd(torch::Tensor x);
torch::Tensor swish_bwd(torch::Tensor g, torch::Tensor x);
torch::Tensor mixup_fwd(torch::Tensor x, torch::Tensor perm, float lam);
torch::Tensor pad_add(torch::Tensor x, torch::Tensor shortcut);
std::vector<torch::Tensor> label_smooth_ce_fwd(torch::Tensor logits, torch::Tensor target, double eps);
torch::Tensor label_smooth_ce_bwd(torch::Tensor softmax, torch::Tensor target,
                                  torch::Tensor grad_loss, double eps);
void sgd_fused_step(torch::Tensor p, torch::Tensor g, torch::Tensor buf,
                    torch::Tensor normsq, torch::Tensor lr_t, int64_t n_decay,
                    double wd, double clip, double momentum, int64_t nesterov);
void sgd_fused_step_mixed(torch::Tensor master, torch::Tensor work, torch::Tensor g,
                          torch::Tensor buf, torch::Tensor normsq, torch::Tensor lr_t,
                          int64_t n_decay, double wd, double clip, double momentum,
                          int64_t nesterov);
void ema_lerp_(torch::Tensor shadow, torch::Tensor x, double mu);
void gather_grads(torch::Tensor table, torch::Tensor flat);
void rmsprop_fused_step_mixed(torch::Tensor master, torch::Tensor work, torch::Tensor g,
                              torch::Tensor ms, torch::Tensor mom, torch::Tensor normsq,
                              torch::Tensor lr_t, int64_t n_decay, double wd,
                              double clip, double rho, double momentum, double eps);
torch::Tensor aug_pipeline(torch::Tensor images, torch::Tensor sel, torch::Tensor prog,
                           torch::Tensor post, torch::Tensor mean, torch::Tensor std,
                           bool bf16_out);
torch::Tensor aug_pipeline_imagenet(torch::Tensor images, torch::Tensor sel,
                                    torch::Tensor prog, torch::Tensor post,
                                    torch::Tensor mean, torch::Tensor std,
                                    int64_t out_h, int64_t out_w, bool bf16_out);
std::vector<torch::Tensor> bn_relu_fwd(torch::Tensor x, torch::Tensor gamma,
                                       torch::Tensor beta, torch::Tensor running_mean,
                                       torch::Tensor running_var, bool training,
                                       double momentum, double eps, int64_t act,
                                       torch::Tensor pre_scratch);
std::vector<torch::Tensor> residual_add_bn_stats(torch::Tensor a, torch::Tensor b);
std::vector<torch::Tensor> bn_relu_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor out, torch::Tensor mean,
                                       torch::Tensor invstd, torch::Tensor gamma,
                                       torch::Tensor beta, bool training, int64_t act);
torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         int64_t stride, int64_t pad);
torch::Tensor conv2d_bwd_data(torch::Tensor dy, torch::Tensor w, int64_t stride,
                              int64_t pad, int64_t H, int64_t W_in);
torch::Tensor conv2d_bwd_data_s2(torch::Tensor dy, torch::Tensor w,
                                 int64_t H, int64_t W, bool pre_flipped);
torch::Tensor conv2d_fwd_grouped(torch::Tensor x, torch::Tensor w,
                                 torch::Tensor bias, int64_t stride,
                                 int64_t pad, int64_t groups);
std::vector<torch::Tensor> conv2d_bwd_weight(torch::Tensor dy, torch::Tensor x,
                                             int64_t stride, int64_t pad,
                                             int64_t KH, int64_t KW, bool want_bias);
torch::Tensor colsum_bf16(torch::Tensor dy);
torch::Tensor colsum_bf16_legacy(torch::Tensor dy);
void colsum_bf16_ws(torch::Tensor dy, torch::Tensor part, torch::Tensor out);
void flip_weights_batched(torch::Tensor table, int64_t total);
torch::Tensor dwconv_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         int64_t stride, int64_t pt, int64_t pb, int64_t pl, int64_t pr);
torch::Tensor dwconv_bwd_data(torch::Tensor dy, torch::Tensor w, int64_t stride,
                              int64_t pt, int64_t pl, int64_t H, int64_t W);
torch::Tensor dwconv_bwd_weight(torch::Tensor dy, torch::Tensor x, int64_t stride,
                                int64_t pt, int64_t pl, int64_t KH, int64_t KW);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("scale_bcast", &scale_bcast, "out = x * s[b] (per-sample broadcast)");
  m.def("scale_lerp", &scale_lerp, "out = a[b]*x1 + (1-a[b])*x2");
  m.def("swish_fwd", &swish_fwd);
  m.def("swish_bwd", &swish_bwd);
  m.def("mixup_fwd", &mixup_fwd);
  m.def("pad_add", &pad_add, "out = x + zero-channel-padded shortcut (NHWC)");
  m.def("label_smooth_ce_fwd", &label_smooth_ce_fwd);
  m.def("label_smooth_ce_bwd", &label_smooth_ce_bwd);
  m.def("sgd_fused_step", &sgd_fused_step,
        "fused manual-WD + global-clip + nesterov SGD on flat buffers");
  m.def("sgd_fused_step_mixed", &sgd_fused_step_mixed,
        "mixed bf16-work/fp32-master fused SGD step");
  m.def("ema_lerp_", &ema_lerp_);
  m.def("rmsprop_fused_step_mixed", &rmsprop_fused_step_mixed,
        "fused TF-semantics RMSprop on flat mixed-precision buffers");
  m.def("gather_grads", &gather_grads,
        "pack scattered autograd grads into the flat bf16 buffer");
  m.def("aug_pipeline_imagenet", &aug_pipeline_imagenet,
        "imagenet pipeline: program ops + EffNet box crop-resize + jitter + lighting");
  m.def("aug_pipeline", &aug_pipeline,
        "batched augmentation program executor (uint8 NHWC -> normalized bf16/f32)");
  m.def("bn_relu_fwd", &bn_relu_fwd);
  m.def("residual_add_bn_stats", &residual_add_bn_stats,
        "out = a + b, plus BN fwd-reduce partials for the following bn");
  m.def("conv2d_fwd", &conv2d_fwd, "MFMA implicit-GEMM NHWC bf16 conv forward");
  m.def("conv2d_bwd_data", &conv2d_bwd_data);
  m.def("conv2d_bwd_data_s2", &conv2d_bwd_data_s2,
        "stride-2 3x3 bwd-data via phase decomposition");
  m.def("conv2d_fwd_grouped", &conv2d_fwd_grouped,
        "grouped 3x3 s1 conv (cardinality branches) on the direct kernel");
  m.def("conv2d_bwd_weight", &conv2d_bwd_weight);
  m.def("colsum_bf16", &colsum_bf16, "channel column-sum (bias grad), replay-safe v2");
  m.def("colsum_bf16_legacy", &colsum_bf16_legacy,
        "round-1 atomic colsum, kept for the hipGraph corruption bisect only");
  m.def("colsum_bf16_ws", &colsum_bf16_ws,
        "colsum v2 into caller-provided workspace/out (graph bisect)");
  m.def("flip_weights_batched", &flip_weights_batched,
        "one-launch bwd-data weight repack for every registered conv");
  m.def("dwconv_fwd", &dwconv_fwd, "depthwise conv forward (NHWC bf16)");
  m.def("dwconv_bwd_data", &dwconv_bwd_data);
  m.def("dwconv_bwd_weight", &dwconv_bwd_weight);
  m.def("bn_relu_bwd", &bn_relu_bwd);
}
