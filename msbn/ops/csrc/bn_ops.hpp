// Host-side API of the msbn gfx950 BatchNorm kernel set (bn_kernels.hip).
// Semantics: SURVEY.md §2.3 (the five stock SyncBatchNorm ATen ops), re-designed
// for MI355X (two-stage deterministic fp64 reductions sized for 256 CUs,
// NCHW + channels-last layouts, packed stat buffers for single-collective sync).
#pragma once

#include <ATen/ATen.h>

#include <tuple>

namespace msbn {

// (mean, invstd) fp32 [C]; invstd = 1/sqrt(biased_var + eps).
std::tuple<at::Tensor, at::Tensor> batch_norm_stats(const at::Tensor& input,
                                                    double eps);

// Fused: write [mean(C) | invstd(C) | count(1)] into `out` (fp32, 2C+1).
void batch_norm_stats_packed(const at::Tensor& input, double eps,
                             at::Tensor& out);

// Combine per-rank stats; counts-weighted Chan merge; in-place running update.
std::tuple<at::Tensor, at::Tensor> batch_norm_gather_stats_with_counts(
    const at::Tensor& mean_all, const at::Tensor& invstd_all,
    const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum, double eps,
    const at::Tensor& counts);

// Packed variant: packed_all is [W, 2C+1]; returns (mean, invstd, count_sum[1]).
// Zero-count ranks masked in-kernel (no GPU->CPU sync; hipGraph-safe).
std::tuple<at::Tensor, at::Tensor, at::Tensor> batch_norm_gather_stats_packed(
    const at::Tensor& packed_all, const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum, double eps);

// As above but ALSO emits the per-channel [scale | shift] coefs (one launch).
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_gather_stats_packed_coefs(
    const at::Tensor& packed_all, const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum, double eps,
    const c10::optional<at::Tensor>& weight,
    const c10::optional<at::Tensor>& bias, bool want_coefs);

// Fused local (world_size==1) stats: running-stats update + coefs emitted by
// the finalize kernel -> (mean, invstd, count_sum[1], coefs-or-undefined).
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_stats_local(const at::Tensor& input, double eps,
                       const c10::optional<at::Tensor>& running_mean,
                       const c10::optional<at::Tensor>& running_var,
                       double momentum,
                       const c10::optional<at::Tensor>& weight,
                       const c10::optional<at::Tensor>& bias, bool want_coefs);

// y = (x - mean) * invstd * weight + bias (scale/shift precomputed per channel).
at::Tensor batch_norm_elemt(const at::Tensor& input,
                            const c10::optional<at::Tensor>& weight,
                            const c10::optional<at::Tensor>& bias,
                            const at::Tensor& mean, const at::Tensor& invstd,
                            double eps);

// (sum_dy, sum_dy_xmu, grad_weight, grad_bias); sum_dy / sum_dy_xmu are views
// into ONE contiguous [2C] fp32 buffer so the cross-rank all_reduce is a single
// message (S6) with no cat() kernel.
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_backward_reduce(const at::Tensor& grad_out, const at::Tensor& input,
                           const at::Tensor& mean, const at::Tensor& invstd,
                           const c10::optional<at::Tensor>& weight, bool input_g,
                           bool weight_g, bool bias_g);

// grad_input; `count` is a device tensor: [1] fp32 total count, or [W]
// per-rank counts (any int/float dtype) which are summed on device.
at::Tensor batch_norm_backward_elemt(
    const at::Tensor& grad_out, const at::Tensor& input, const at::Tensor& mean,
    const at::Tensor& invstd, const c10::optional<at::Tensor>& weight,
    const at::Tensor& sum_dy, const at::Tensor& sum_dy_xmu,
    const at::Tensor& count);

// ---- fused BN(+residual add)(+ReLU) epilogue path -------------------------
// Forward: y = relu?(x*scale + shift [+ residual]).  Backward recomputes the
// ReLU gate in-kernel from (x, scale, shift[, residual]) — no mask tensor and
// no separate clamp/add/threshold_backward kernels or their full-tensor
// round trips.
// coefs = packed [scale(C) | shift(C)] fp32 (bn_make_coefs); pass it to the
// act ops to avoid recomputing the per-channel affine in each of them.
at::Tensor bn_make_coefs(const at::Tensor& mean, const at::Tensor& invstd,
                         const c10::optional<at::Tensor>& weight,
                         const c10::optional<at::Tensor>& bias);

at::Tensor batch_norm_elemt_act(const at::Tensor& input,
                                const c10::optional<at::Tensor>& residual,
                                const c10::optional<at::Tensor>& weight,
                                const c10::optional<at::Tensor>& bias,
                                const at::Tensor& mean,
                                const at::Tensor& invstd, bool relu,
                                const c10::optional<at::Tensor>& coefs_in);

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_backward_reduce_act(
    const at::Tensor& grad_out, const at::Tensor& input,
    const c10::optional<at::Tensor>& residual, const at::Tensor& mean,
    const at::Tensor& invstd, const c10::optional<at::Tensor>& weight,
    const c10::optional<at::Tensor>& bias, bool relu_mask, bool input_g,
    bool weight_g, bool bias_g, const c10::optional<at::Tensor>& coefs_in,
    const c10::optional<at::Tensor>& gm_out);

// ---- small-plane world-1 fused path (single launch fwd / bwd) -------------
// Stock K10-style block-per-channel kernels gated to small NCHW planes
// (GAN / small-per-GPU-batch regime): one kernel does stats + running
// update + normalize(+res)(+relu); one kernel does the whole backward.
bool bn_fused_local_eligible(const at::Tensor& input,
                             const c10::optional<at::Tensor>& weight,
                             const c10::optional<at::Tensor>& bias,
                             const c10::optional<at::Tensor>& running_mean,
                             const c10::optional<at::Tensor>& running_var);

// returns (y, mean, invstd, count[1], coefs[2C])
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_fwd_fused_local(const at::Tensor& input,
                           const c10::optional<at::Tensor>& residual,
                           const c10::optional<at::Tensor>& weight,
                           const c10::optional<at::Tensor>& bias, double eps,
                           double momentum,
                           const c10::optional<at::Tensor>& running_mean,
                           const c10::optional<at::Tensor>& running_var,
                           bool relu);

// returns (dx, grad_weight?, grad_bias?, dres?)
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_bwd_fused_local(const at::Tensor& grad_out, const at::Tensor& input,
                           const c10::optional<at::Tensor>& residual,
                           const at::Tensor& mean, const at::Tensor& invstd,
                           const c10::optional<at::Tensor>& weight,
                           const c10::optional<at::Tensor>& coefs,
                           bool relu_mask, bool want_res_grad, bool weight_g,
                           bool bias_g);

// returns (grad_input, grad_residual-or-undefined)
std::tuple<at::Tensor, at::Tensor> batch_norm_backward_elemt_act(
    const at::Tensor& grad_out, const at::Tensor& input,
    const c10::optional<at::Tensor>& residual, const at::Tensor& mean,
    const at::Tensor& invstd, const c10::optional<at::Tensor>& weight,
    const c10::optional<at::Tensor>& bias, const at::Tensor& sum_dy,
    const at::Tensor& sum_dy_xmu, const at::Tensor& count, bool relu_mask,
    bool want_res_grad, const c10::optional<at::Tensor>& coefs_in);

}  // namespace msbn
