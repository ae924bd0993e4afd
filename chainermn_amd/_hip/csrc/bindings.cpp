// pybind11 bindings for chainermn_amd._C (gfx950 kernels).

#include <torch/extension.h>

void batched_pack(torch::Tensor ptrs, torch::Tensor sizes,
                  torch::Tensor offsets, torch::Tensor out,
                  torch::ScalarType src_dtype);
void batched_unpack(torch::Tensor ptrs, torch::Tensor sizes,
                    torch::Tensor offsets, torch::Tensor in, double scale,
                    torch::ScalarType dst_dtype);
void cast_flat(torch::Tensor src, torch::Tensor dst, double scale);
void fused_sgd(torch::Tensor p, torch::Tensor p_lowp, torch::Tensor g,
               torch::Tensor v, double lr, double momentum, double wd,
               double dampening, bool nesterov, double gscale);
void fused_adam(torch::Tensor p, torch::Tensor p_lowp, torch::Tensor g,
                torch::Tensor m, torch::Tensor v, double lr, double b1,
                double b2, double eps, double wd, long step, bool adamw,
                double gscale);
void segment_sq_norms(torch::Tensor flat, torch::Tensor offsets,
                      torch::Tensor sizes, torch::Tensor out);
void lars_update(torch::Tensor p, torch::Tensor p_lowp, torch::Tensor g,
                 torch::Tensor v, torch::Tensor offsets, torch::Tensor sizes,
                 torch::Tensor p_sqn, torch::Tensor g_sqn, double lr,
                 double momentum, double wd, double eta, double eps,
                 double gscale);
std::tuple<torch::Tensor, torch::Tensor, long> bn_stats(torch::Tensor x);
void bn_stats_into(torch::Tensor x, torch::Tensor buf);
void bn_bwd_reduce_into(torch::Tensor x, torch::Tensor gy,
                        torch::Tensor mean, torch::Tensor rstd,
                        torch::Tensor buf);
void bn_bwd_reduce_act_into(torch::Tensor x, torch::Tensor gy,
                            torch::Tensor mean, torch::Tensor rstd,
                            torch::Tensor gamma, torch::Tensor beta,
                            torch::Tensor yout, long relu_mode,
                            torch::Tensor buf);
torch::Tensor bn_fwd_act(torch::Tensor x, torch::Tensor mean,
                         torch::Tensor rstd, torch::Tensor gamma,
                         torch::Tensor beta, torch::Tensor res,
                         long relu_mode);
std::tuple<torch::Tensor, torch::Tensor> bn_bwd_input_act(
    torch::Tensor x, torch::Tensor gy, torch::Tensor mean,
    torch::Tensor rstd, torch::Tensor gamma, torch::Tensor beta,
    torch::Tensor gbeta, torch::Tensor ggamma, torch::Tensor yout,
    long relu_mode, bool want_gres, double total_count);
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> bn_finalize(
    torch::Tensor buf, double count, double eps, double momentum,
    double unbias, torch::Tensor r_mean, torch::Tensor r_var);
std::tuple<torch::Tensor, torch::Tensor> bn_bwd_reduce(torch::Tensor x,
                                                       torch::Tensor gy,
                                                       torch::Tensor mean,
                                                       torch::Tensor rstd);
torch::Tensor bn_fwd(torch::Tensor x, torch::Tensor mean, torch::Tensor rstd,
                     torch::Tensor gamma, torch::Tensor beta);
std::tuple<torch::Tensor, torch::Tensor> maxpool3x3s2_fwd(torch::Tensor x);
torch::Tensor gemm_bt_bf16(torch::Tensor a, torch::Tensor b);
torch::Tensor gemm_bt256_bf16(torch::Tensor a, torch::Tensor b);
torch::Tensor gemm_bt256p_bf16(torch::Tensor a, torch::Tensor b,
                               long sched);
torch::Tensor gemm_bt256p_stats_bf16(torch::Tensor a, torch::Tensor b,
                                     torch::Tensor stats, long sched);
torch::Tensor conv3x3s1_fwd_bf16(torch::Tensor x, torch::Tensor w,
                                 torch::Tensor stats);
torch::Tensor conv3x3_wgrad_bf16(torch::Tensor x, torch::Tensor gy);
torch::Tensor gemm_btw_bf16(torch::Tensor a, torch::Tensor b,
                            torch::Tensor stats);
torch::Tensor gemm_btw_acc_bf16(torch::Tensor a, torch::Tensor b,
                                torch::Tensor out);
torch::Tensor gemm_bt256p_acc_bf16(torch::Tensor a, torch::Tensor b,
                                   torch::Tensor out);
torch::Tensor gemm_bt256p_bnbwd_bf16(
    torch::Tensor a, torch::Tensor b, torch::Tensor bn_x,
    torch::Tensor bn_mean, torch::Tensor bn_rstd, torch::Tensor bn_gamma,
    torch::Tensor bn_beta, torch::Tensor bn_partials);
torch::Tensor gemm_btw_bnbwd_bf16(torch::Tensor a, torch::Tensor b,
                                  torch::Tensor bn_x,
                                  torch::Tensor bn_mean,
                                  torch::Tensor bn_rstd,
                                  torch::Tensor bn_gamma,
                                  torch::Tensor bn_beta,
                                  torch::Tensor bn_partials);
torch::Tensor conv3x3s1_fwd_bnbwd_bf16(
    torch::Tensor x, torch::Tensor w, torch::Tensor stats,
    torch::Tensor bn_x, torch::Tensor bn_mean, torch::Tensor bn_rstd,
    torch::Tensor bn_gamma, torch::Tensor bn_beta,
    torch::Tensor bn_partials);
torch::Tensor maxpool3x3s2_bwd(torch::Tensor gy, torch::Tensor idx,
                               long IH, long IW);
torch::Tensor bn_bwd_input(torch::Tensor x, torch::Tensor gy,
                           torch::Tensor mean, torch::Tensor rstd,
                           torch::Tensor gamma, torch::Tensor gbeta,
                           torch::Tensor ggamma, double total_count);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "chainermn_amd MI355X (gfx950) kernels";
  m.def("batched_pack", &batched_pack,
        "gather param grads into flat wire buffer (fused cast, K1)");
  m.def("batched_unpack", &batched_unpack,
        "scatter flat buffer to grads (fused scale+cast, K2)");
  m.def("cast_flat", &cast_flat, "flat cast with fused scale (K3)");
  m.def("fused_sgd", &fused_sgd, "flat-arena SGD/momentum step (K5)");
  m.def("fused_adam", &fused_adam, "flat-arena Adam/AdamW step (K5)");
  m.def("segment_sq_norms", &segment_sq_norms,
        "per-layer sum of squares (LARS)");
  m.def("lars_update", &lars_update, "per-layer trust-ratio update (LARS)");
  m.def("bn_stats", &bn_stats, "per-channel (sum, sumsq, count) (K6)");
  m.def("bn_stats_into", &bn_stats_into,
        "per-channel (sum|sumsq) into a caller buffer (K6)");
  m.def("bn_bwd_reduce_into", &bn_bwd_reduce_into,
        "per-channel (sum gy|sum gy*xhat) into a caller buffer (K6)");
  m.def("bn_finalize", &bn_finalize,
        "mean/rstd/var from the reduced stats buffer, optional fused "
        "running-stat update (one launch)",
        py::arg("buf"), py::arg("count"), py::arg("eps"),
        py::arg("momentum") = 0.0, py::arg("unbias") = 1.0,
        py::arg("r_mean") = torch::Tensor(),
        py::arg("r_var") = torch::Tensor());
  m.def("bn_fwd", &bn_fwd, "BN normalize with global stats (K7)");
  m.def("bn_bwd_reduce", &bn_bwd_reduce,
        "per-channel (sum gy, sum gy*xhat) (K6)");
  m.def("bn_bwd_input", &bn_bwd_input, "BN input gradient (K7)");
  m.def("bn_bwd_reduce_act_into", &bn_bwd_reduce_act_into,
        "BN backward reduce with fused relu mask");
  m.def("bn_fwd_act", &bn_fwd_act,
        "y = act(bn(x) [+ res]) (fused relu / residual add)");
  m.def("bn_bwd_input_act", &bn_bwd_input_act,
        "BN input grad with fused relu mask [+ gres]");
  m.def("maxpool3x3s2_fwd", &maxpool3x3s2_fwd,
        "channels-last 3x3/s2/p1 maxpool forward (+window codes)");
  m.def("maxpool3x3s2_bwd", &maxpool3x3s2_bwd,
        "gather-based maxpool backward (no atomics)");
  m.def("conv3x3s1_fwd_bf16",
        [](torch::Tensor x, torch::Tensor w,
           c10::optional<torch::Tensor> stats) {
          return conv3x3s1_fwd_bf16(
              x, w, stats.has_value() ? *stats : torch::Tensor());
        },
        "implicit-GEMM 3x3/s1/p1 conv forward (phased MFMA); optional "
        "fused BN-stats partials [16, 2*Cout]",
        py::arg("x"), py::arg("w"), py::arg("stats") = py::none());
  m.def("conv3x3_wgrad_bf16", &conv3x3_wgrad_bf16,
        "implicit-GEMM 3x3/s1/p1 conv weight-grad (pair-transposed LDS "
        "staging, split-M fp32 atomics)");
  m.def("gemm_btw_bf16",
        [](torch::Tensor a, torch::Tensor b,
           c10::optional<torch::Tensor> stats) {
          return gemm_btw_bf16(
              a, b, stats.has_value() ? *stats : torch::Tensor());
        },
        "narrow-N phased MFMA GEMM (BN 128/64): D = A @ B^T; optional "
        "fused BN-stats partials",
        py::arg("a"), py::arg("b"), py::arg("stats") = py::none());
  m.def("gemm_bt256p_bnbwd_bf16", &gemm_bt256p_bnbwd_bf16,
        "256-tile phased GEMM with the produced-gy BN-backward reduce "
        "fused into the epilogue");
  m.def("gemm_btw_bnbwd_bf16", &gemm_btw_bnbwd_bf16,
        "narrow-N phased GEMM with fused BN-backward reduce");
  m.def("conv3x3s1_fwd_bnbwd_bf16", &conv3x3s1_fwd_bnbwd_bf16,
        "conv3x3 (dgrad form) with fused BN-backward reduce");
  m.def("gemm_btw_acc_bf16", &gemm_btw_acc_bf16,
        "narrow-N phased GEMM, accumulate mode: out += A @ B^T");
  m.def("gemm_bt256p_acc_bf16", &gemm_bt256p_acc_bf16,
        "256-tile phased GEMM, accumulate mode: out += A @ B^T");
  m.def("gemm_bt256p_stats_bf16", &gemm_bt256p_stats_bf16,
        "256-tile phased GEMM with fused per-column BN-stats partials",
        py::arg("a"), py::arg("b"), py::arg("stats"),
        py::arg("sched") = 3);
  m.def("gemm_bt_bf16", &gemm_bt_bf16,
        "hand-written MFMA bf16 GEMM: D = A @ B^T (128x128x64 tile, "
        "glds staging, XOR-swizzled LDS, XCD-aware grid)");
  m.def("gemm_bt256_bf16", &gemm_bt256_bf16,
        "256x256x64-tile 8-wave MFMA bf16 GEMM: D = A @ B^T");
  m.def("gemm_bt256p_bf16", &gemm_bt256p_bf16,
        "256-tile GEMM, 8-phase schedule (counted vmcnt, raw barriers, "
        "setprio): D = A @ B^T.  sched bit0 = stage-before-reads, "
        "bit1 = single barrier per phase",
        py::arg("a"), py::arg("b"), py::arg("sched") = 0);
}
