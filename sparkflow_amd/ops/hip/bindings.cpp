// Python bindings for the sparkflow_amd gfx950 kernel extension.
#include <torch/extension.h>

#include <vector>

// gemm.hip
torch::Tensor gemm_act(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                       int64_t act);
torch::Tensor gemm_dgrad(torch::Tensor dy, torch::Tensor w);
torch::Tensor gemm_dgrad_act(torch::Tensor dy, torch::Tensor y,
                             torch::Tensor w, int64_t act);
torch::Tensor gemm_wgrad(torch::Tensor x, torch::Tensor dy);
torch::Tensor gemm_wgrad_bias(torch::Tensor x, torch::Tensor dy);
torch::Tensor gemm_wgrad_bias_act(torch::Tensor x, torch::Tensor dy,
                                  torch::Tensor y, int64_t act);
torch::Tensor gemm_wgrad_bias_ws(torch::Tensor x, torch::Tensor dy);
torch::Tensor gemm_wgrad_bias_act_ws(torch::Tensor x, torch::Tensor dy,
                                     torch::Tensor y, int64_t act);
torch::Tensor bias_grad(torch::Tensor dy);

// elementwise.hip
torch::Tensor act_bwd(torch::Tensor y, torch::Tensor dy, int64_t act);
torch::Tensor argmax_rows(torch::Tensor x);
void gather_batch(torch::Tensor x, torch::Tensor xb, torch::Tensor idx_pool,
                  torch::Tensor ctr, torch::Tensor done,
                  c10::optional<torch::Tensor> y,
                  c10::optional<torch::Tensor> yb,
                  c10::optional<torch::Tensor> wsrc,
                  c10::optional<torch::Tensor> wdst);
void pack_segments(std::vector<torch::Tensor> srcs, torch::Tensor flat);

// losses.hip
std::vector<torch::Tensor> softmax_xent_fwd(torch::Tensor logits,
                                            torch::Tensor onehot,
                                            bool want_value);
torch::Tensor softmax_xent_bwd(torch::Tensor probs, torch::Tensor onehot,
                               torch::Tensor gscale, double mult);
std::vector<torch::Tensor> softmax_xent_idx_fwd(torch::Tensor logits,
                                                torch::Tensor labels,
                                                bool want_value,
                                                bool dlogits_mode);
torch::Tensor softmax_xent_idx_bwd(torch::Tensor probs, torch::Tensor labels,
                                   torch::Tensor gscale, double mult);
torch::Tensor mse_fwd(torch::Tensor pred, torch::Tensor target,
                      bool want_value);
torch::Tensor mse_bwd(torch::Tensor pred, torch::Tensor target,
                      torch::Tensor gscale, double mult);

// optim.hip
void fused_sgd(torch::Tensor p, torch::Tensor g, double lr);
void fused_momentum(torch::Tensor p, torch::Tensor g, torch::Tensor acc,
                    double lr, double mom, bool nesterov);
void fused_adam(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, double lr_t, double b1, double b2,
                double eps);
void fused_adam_dev(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                    torch::Tensor v, torch::Tensor state, double lr,
                    double b1, double b2, double eps);
void fused_rmsprop(torch::Tensor p, torch::Tensor g, torch::Tensor ms,
                   torch::Tensor mom, double lr, double decay,
                   double momentum, double eps);
void reduce_apply_adam(std::vector<torch::Tensor> ws,
                       std::vector<int64_t> offsets, torch::Tensor p,
                       torch::Tensor m, torch::Tensor v, torch::Tensor state,
                       double lr, double b1, double b2, double eps);

// conv.hip
torch::Tensor conv2d_act(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         int64_t stride, int64_t pad, int64_t act);
torch::Tensor conv_gemm_fwd(torch::Tensor x, torch::Tensor w,
                            torch::Tensor bias, int64_t stride, int64_t pad,
                            int64_t act);
torch::Tensor conv_gemm_wgrad(torch::Tensor x, torch::Tensor dy, int64_t Kh,
                              int64_t Kw, int64_t stride, int64_t pad);
torch::Tensor conv2d_dgrad(torch::Tensor dy, torch::Tensor w, int64_t H,
                           int64_t W, int64_t stride, int64_t pad);
torch::Tensor conv2d_wgrad(torch::Tensor x, torch::Tensor dy, int64_t Kh,
                           int64_t Kw, int64_t stride, int64_t pad);
torch::Tensor conv_bias_grad(torch::Tensor dy);
torch::Tensor conv_dilate(torch::Tensor dy, int64_t s, int64_t extra_h,
                          int64_t extra_w);
std::vector<torch::Tensor> maxpool2d_fwd(torch::Tensor x, int64_t k,
                                         int64_t stride, int64_t pad);
torch::Tensor maxpool2d_bwd(torch::Tensor dy, torch::Tensor idx, int64_t H,
                            int64_t W);

// conv_nhwc.hip — channel-last implicit-GEMM conv family
torch::Tensor conv_nhwc_fwd(torch::Tensor x, torch::Tensor w,
                            torch::Tensor bias, int64_t stride, int64_t pad,
                            int64_t act);
torch::Tensor conv_nhwc_wgrad(torch::Tensor x, torch::Tensor dy, int64_t Kh,
                              int64_t Kw, int64_t stride, int64_t pad);
torch::Tensor conv_nhwc_dilate(torch::Tensor dy, int64_t s, int64_t extra_h,
                               int64_t extra_w);
torch::Tensor conv_nhwc_bias_grad(torch::Tensor dy);
std::vector<torch::Tensor> maxpool_nhwc_fwd(torch::Tensor x, int64_t k,
                                            int64_t stride, int64_t pad);
torch::Tensor maxpool_nhwc_bwd(torch::Tensor dy, torch::Tensor idx,
                               int64_t H, int64_t W, int64_t k,
                               int64_t stride, int64_t pad);
torch::Tensor conv_nhwc_dgrad_strided(torch::Tensor dy, torch::Tensor wt,
                                      int64_t H, int64_t W, int64_t stride,
                                      int64_t pad);

// norm.hip — training-mode BatchNorm2d (+ fused ReLU)
std::vector<torch::Tensor> bn2d_fwd(torch::Tensor x, torch::Tensor gamma,
                                    torch::Tensor beta,
                                    torch::Tensor run_mean,
                                    torch::Tensor run_var, double momentum,
                                    double eps, bool relu);
std::vector<torch::Tensor> bn2d_bwd(torch::Tensor x, torch::Tensor dy,
                                    torch::Tensor y, torch::Tensor gamma,
                                    torch::Tensor save_mean,
                                    torch::Tensor save_invstd, bool relu);
std::vector<torch::Tensor> bn2d_nhwc_fwd(torch::Tensor x,
                                         torch::Tensor gamma,
                                         torch::Tensor beta,
                                         torch::Tensor run_mean,
                                         torch::Tensor run_var,
                                         double momentum, double eps,
                                         bool relu, torch::Tensor res);
std::vector<torch::Tensor> bn2d_nhwc_bwd(torch::Tensor x, torch::Tensor dy,
                                         torch::Tensor y,
                                         torch::Tensor gamma,
                                         torch::Tensor save_mean,
                                         torch::Tensor save_invstd,
                                         bool relu, bool want_dres);

// dist_session.hip — native RCCL session loops (GIL-free)
py::bytes nccl_unique_id();
int64_t nccl_comm_init(py::bytes id_bytes, int64_t nranks, int64_t rank,
                       int64_t timeout_ms);
void nccl_comm_destroy(int64_t h);
void ps_session(int64_t comm_h, int64_t peer, torch::Tensor weights,
                torch::Tensor grad_buf, int64_t k, int64_t opt_id,
                std::vector<torch::Tensor> state, double lr, double h1,
                double h2, double h3);
void worker_session(int64_t comm_h, int64_t peer, torch::Tensor w_flat,
                    torch::Tensor g_flat, int64_t graph_exec, int64_t k);
void raw_send(int64_t comm_h, int64_t peer, torch::Tensor t);
void raw_recv(int64_t comm_h, int64_t peer, torch::Tensor t);
void multi_replay(std::vector<int64_t> execs, int64_t k);
void replay_streams_init(int64_t n);
void replay_streams_reroll();

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "sparkflow_amd gfx950 (CDNA4) kernels";
  m.def("gemm_act", &gemm_act, "fused fp32 MFMA GEMM + bias + activation");
  m.def("gemm_dgrad", &gemm_dgrad, "dx = dy @ W^T");
  m.def("gemm_dgrad_act", &gemm_dgrad_act,
        "dx = act_grad(y, dy) @ W^T (act bwd fused into the A load)");
  m.def("gemm_wgrad", &gemm_wgrad, "dW = x^T @ dy");
  m.def("gemm_wgrad_bias", &gemm_wgrad_bias,
        "dWb = [x;ones]^T @ dy (bias grad fused as last row)");
  m.def("gemm_wgrad_bias_act", &gemm_wgrad_bias_act,
        "dWb = [x;ones]^T @ act_grad(y, dy) (act bwd fused into B load)");
  m.def("gemm_wgrad_bias_ws", &gemm_wgrad_bias_ws,
        "wgrad+bias split-K workspace, reduce deferred");
  m.def("gemm_wgrad_bias_act_ws", &gemm_wgrad_bias_act_ws,
        "wgrad+bias+actbwd split-K workspace, reduce deferred");
  m.def("bias_grad", &bias_grad, "column-sum bias gradient");
  m.def("act_bwd", &act_bwd, "activation backward from output");
  m.def("argmax_rows", &argmax_rows, "rowwise argmax head");
  m.def("gather_batch", &gather_batch,
        "fused minibatch gather + pool-cursor advance (hipGraph step "
        "prologue); optionally also performs the weight-pull copy",
        py::arg("x"), py::arg("xb"), py::arg("idx_pool"),
        py::arg("ctr"), py::arg("done"), py::arg("y") = py::none(),
        py::arg("yb") = py::none(), py::arg("wsrc") = py::none(),
        py::arg("wdst") = py::none());
  m.def("pack_segments", &pack_segments,
        "pack <=16 grad tensors into the flat push buffer in one launch");
  m.def("softmax_xent_fwd", &softmax_xent_fwd, "fused softmax+xent fwd",
        py::arg("logits"), py::arg("onehot"), py::arg("want_value") = true);
  m.def("softmax_xent_bwd", &softmax_xent_bwd, "softmax+xent bwd");
  m.def("softmax_xent_idx_fwd", &softmax_xent_idx_fwd,
        "fused softmax+xent fwd, class-index labels (int64 or fp32); "
        "dlogits_mode fuses the backward into the forward",
        py::arg("logits"), py::arg("labels"), py::arg("want_value") = true,
        py::arg("dlogits_mode") = false);
  m.def("softmax_xent_idx_bwd", &softmax_xent_idx_bwd,
        "softmax+xent bwd, class-index labels");
  m.def("mse_fwd", &mse_fwd, "fused MSE loss", py::arg("pred"),
        py::arg("target"), py::arg("want_value") = true);
  m.def("mse_bwd", &mse_bwd, "MSE grad");
  m.def("fused_sgd", &fused_sgd, "single-launch SGD apply");
  m.def("fused_momentum", &fused_momentum, "single-launch momentum apply");
  m.def("fused_adam", &fused_adam, "single-launch Adam apply");
  m.def("fused_adam_dev", &fused_adam_dev,
        "Adam apply with device-resident step state (hipGraph-safe)");
  m.def("fused_rmsprop", &fused_rmsprop, "single-launch RMSProp apply");
  m.def("reduce_apply_adam", &reduce_apply_adam,
        "sum per-layer wgrad workspaces + device-state Adam apply in one "
        "launch (replaces split-K reduces + pack_segments + adam)");
  m.def("conv2d_act", &conv2d_act, "conv2d + bias + activation fwd (direct)");
  m.def("conv_gemm_fwd", &conv_gemm_fwd,
        "implicit-GEMM MFMA conv2d + bias + activation fwd");
  m.def("conv_gemm_wgrad", &conv_gemm_wgrad,
        "implicit-GEMM MFMA conv2d weight grad (split-K)");
  m.def("conv2d_dgrad", &conv2d_dgrad, "conv2d input grad");
  m.def("conv2d_wgrad", &conv2d_wgrad, "conv2d weight grad");
  m.def("conv_bias_grad", &conv_bias_grad, "conv bias grad");
  m.def("conv_dilate", &conv_dilate,
        "zero-stuff dy by stride (strided dgrad as stride-1 conv)");
  m.def("maxpool2d_fwd", &maxpool2d_fwd, "maxpool fwd with argmax indices");
  m.def("maxpool2d_bwd", &maxpool2d_bwd, "maxpool bwd scatter");
  m.def("conv_nhwc_fwd", &conv_nhwc_fwd,
        "NHWC implicit-GEMM conv fwd (+bias+act); w [Cout,Kh,Kw,Cin]");
  m.def("conv_nhwc_wgrad", &conv_nhwc_wgrad,
        "NHWC implicit-GEMM conv weight grad (split-K)");
  m.def("conv_nhwc_dilate", &conv_nhwc_dilate,
        "zero-stuff NHWC dy by stride (strided dgrad as stride-1 conv)");
  m.def("conv_nhwc_bias_grad", &conv_nhwc_bias_grad,
        "NHWC conv bias grad (channel-contiguous column sums)");
  m.def("maxpool_nhwc_fwd", &maxpool_nhwc_fwd,
        "NHWC maxpool fwd with argmax indices");
  m.def("conv_nhwc_dgrad_strided", &conv_nhwc_dgrad_strided,
        "direct stride>1 NHWC dgrad (s^2 sub-convolutions, no dilation)");
  m.def("maxpool_nhwc_bwd", &maxpool_nhwc_bwd,
        "NHWC maxpool bwd (float4 window-gather; scalar scatter fallback)");
  m.def("bn2d_fwd", &bn2d_fwd,
        "training BatchNorm2d fwd (+ fused ReLU); returns y, mean, invstd");
  m.def("bn2d_bwd", &bn2d_bwd,
        "BatchNorm2d bwd (+ fused ReLU bwd); returns dx, dgamma, dbeta");
  m.def("bn2d_nhwc_fwd", &bn2d_nhwc_fwd,
        "NHWC training BatchNorm2d fwd (float4; optional fused residual"
        " add + relu)");
  m.def("bn2d_nhwc_bwd", &bn2d_nhwc_bwd,
        "NHWC BatchNorm2d bwd (optionally emits d_res = relu-masked dy)");
  m.def("nccl_unique_id", &nccl_unique_id, "RCCL unique id for a pair comm");
  m.def("nccl_comm_init", &nccl_comm_init,
        "init this rank's side of a raw RCCL communicator (non-blocking\n"
        "rendezvous with a bounded wait; aborts the comm on timeout)",
        py::arg("id_bytes"), py::arg("nranks"), py::arg("rank"),
        py::arg("timeout_ms") = 30000);
  m.def("nccl_comm_destroy", &nccl_comm_destroy, "destroy a raw comm");
  m.def("ps_session", &ps_session,
        "GIL-free k-step PS loop: send weights / recv grads / fused apply");
  m.def("worker_session", &worker_session,
        "GIL-free k-step worker loop: recv weights / graph launch / send");
  m.def("raw_send", &raw_send, "one fp32 send on a raw comm");
  m.def("raw_recv", &raw_recv, "one fp32 recv on a raw comm");
  m.def("replay_streams_init", &replay_streams_init,
        "pre-create dedicated replay streams (deterministic HW-queue map)");
  m.def("replay_streams_reroll", &replay_streams_reroll,
        "re-roll replay stream HW-queue placement (device must be idle)");
  m.def("multi_replay", &multi_replay,
        "k round-robin launches of W captured step graphs on W streams");
}
