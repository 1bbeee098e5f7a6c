// Python bindings for the seist_amd CDNA4 (gfx950) kernel library.

#include <torch/extension.h>

at::Tensor pw_conv_fwd(const at::Tensor& x, const at::Tensor& w,
                       const c10::optional<at::Tensor>& bias);
at::Tensor pw_conv_multi_fwd(std::vector<at::Tensor> xs, const at::Tensor& w,
                             const c10::optional<at::Tensor>& bias);
std::vector<at::Tensor> pw_conv_multi_dx(const at::Tensor& dy,
                                         const at::Tensor& w,
                                         std::vector<long> sizes);
std::vector<at::Tensor> pw_conv_bwd(const at::Tensor& dy, const at::Tensor& x,
                                    const at::Tensor& w, bool has_bias);

at::Tensor conv1d_fwd(const at::Tensor& x, const at::Tensor& w,
                      const c10::optional<at::Tensor>& bias, long stride,
                      long padl, long padr, long groups, long dilation);
std::vector<at::Tensor> conv1d_bwd(const at::Tensor& dy, const at::Tensor& x,
                                   const at::Tensor& w, long stride,
                                   long padl, long padr, long groups,
                                   long dilation, bool has_bias);
at::Tensor conv_transpose1d_fwd(const at::Tensor& x, const at::Tensor& w,
                                const c10::optional<at::Tensor>& bias,
                                long stride);
std::vector<at::Tensor> conv_transpose1d_bwd(const at::Tensor& dy,
                                             const at::Tensor& x,
                                             const at::Tensor& w,
                                             long stride, bool has_bias);

std::vector<at::Tensor> bn_act_fwd(const at::Tensor& x, const at::Tensor& gamma,
                                   const at::Tensor& beta,
                                   const c10::optional<at::Tensor>& running_mean,
                                   const c10::optional<at::Tensor>& running_var,
                                   bool training, double momentum, double eps,
                                   long act);
std::vector<at::Tensor> bn_act_bwd(const at::Tensor& dy, const at::Tensor& x,
                                   const at::Tensor& gamma,
                                   const at::Tensor& beta,
                                   const at::Tensor& mean,
                                   const at::Tensor& invstd, bool training,
                                   long act);

at::Tensor bn_sums_only(const at::Tensor& x);
std::vector<at::Tensor> pw_conv_fwd_stats(
    const at::Tensor& x, const at::Tensor& w,
    const c10::optional<at::Tensor>& bias);
std::vector<at::Tensor> conv1d_fwd_stats(
    const at::Tensor& x, const at::Tensor& w,
    const c10::optional<at::Tensor>& bias, long stride, long padl,
    long padr, long groups, long dilation);
std::vector<at::Tensor> bn_act_fwd_with_part(
    const at::Tensor& x, const at::Tensor& part, const at::Tensor& gamma,
    const at::Tensor& beta, const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum, double eps,
    long act);
at::Tensor bn_part_to_sums(const at::Tensor& part);
std::vector<at::Tensor> bn_act_cat_fwd(
    std::vector<at::Tensor> xs, const at::Tensor& gamma,
    const at::Tensor& beta, const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, bool training,
    double momentum, double eps, long act);
std::vector<at::Tensor> bn_act_cat_bwd(
    const at::Tensor& dy, std::vector<at::Tensor> xs,
    const at::Tensor& gamma, const at::Tensor& beta, const at::Tensor& mean,
    const at::Tensor& invstd, bool training, long act);
std::vector<at::Tensor> bn_finalize_only(
    const at::Tensor& sums, double count,
    const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum,
    double eps);
at::Tensor pw_conv_pre_fwd(const at::Tensor& x, const at::Tensor& w,
                           const c10::optional<at::Tensor>& bias,
                           const at::Tensor& scale, const at::Tensor& shift,
                           long act);
at::Tensor pw_conv_dx(const at::Tensor& dy, const at::Tensor& w);
at::Tensor bn_bwd_dx_eval(const at::Tensor& dy, const at::Tensor& x,
                          const at::Tensor& mean, const at::Tensor& invstd,
                          const at::Tensor& gamma, const at::Tensor& beta,
                          long act);
at::Tensor pw_dw_pre(const at::Tensor& dy, const at::Tensor& x,
                     const c10::optional<at::Tensor>& scale,
                     const c10::optional<at::Tensor>& shift, long act,
                     c10::optional<at::ScalarType> out_dtype);
std::vector<at::Tensor> bn_act_fwd_from_sums(
    const at::Tensor& x, const at::Tensor& sums, double count,
    const at::Tensor& gamma, const at::Tensor& beta,
    const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum, double eps,
    long act);
at::Tensor bn_bwd_sums_only(const at::Tensor& dy, const at::Tensor& x,
                            const at::Tensor& mean, const at::Tensor& invstd,
                            const at::Tensor& gamma, const at::Tensor& beta,
                            long act);
at::Tensor bn_bwd_dx_from_sums(const at::Tensor& dy, const at::Tensor& x,
                               const at::Tensor& mean,
                               const at::Tensor& invstd,
                               const at::Tensor& gamma, const at::Tensor& beta,
                               const at::Tensor& sums, double count, long act);

std::vector<at::Tensor> avgmax_pool_fwd(const at::Tensor& x, long k);
std::vector<at::Tensor> max_pool1d_fwd(const at::Tensor& x, long k,
                                       bool ceil_mode);
at::Tensor max_pool1d_bwd(const at::Tensor& dy, const at::Tensor& argmax,
                          long k, long in_len);
at::Tensor gap_fwd(const at::Tensor& x);
at::Tensor gap_bwd(const at::Tensor& dy, long in_len);
at::Tensor loss_sum_fwd(const at::Tensor& p, const at::Tensor& t,
                        const at::Tensor& w, long kind, double inv_div);
at::Tensor loss_sum_bwd(const at::Tensor& p, const at::Tensor& t,
                        const at::Tensor& w, const at::Tensor& gout,
                        long kind, double inv_div);
at::Tensor avgmax_pool_bwd(const at::Tensor& dy, const at::Tensor& argmax,
                           long k, long in_len);
at::Tensor interp_linear_fwd(const at::Tensor& x, long out_len);
at::Tensor upsample2x_fwd(const at::Tensor& x);
at::Tensor upsample2x_bwd(const at::Tensor& dy);
std::vector<at::Tensor> pooled_attn_train_fwd(const at::Tensor& q,
                                              const at::Tensor& k,
                                              const at::Tensor& v, double p);
std::vector<at::Tensor> pooled_attn_bwd(const at::Tensor& q,
                                        const at::Tensor& k,
                                        const at::Tensor& v,
                                        const at::Tensor& out,
                                        const at::Tensor& dout,
                                        const at::Tensor& stats,
                                        const at::Tensor& mask, double p);
at::Tensor pooled_attn_fwd(const at::Tensor& q, const at::Tensor& k,
                           const at::Tensor& v);
at::Tensor interp_linear_bwd(const at::Tensor& dy, long in_len);

std::vector<at::Tensor> ln_fwd(const at::Tensor& x, const at::Tensor& gamma,
                               const at::Tensor& beta, double eps);
std::vector<at::Tensor> ln_bwd(const at::Tensor& dy, const at::Tensor& x,
                               const at::Tensor& gamma,
                               const at::Tensor& mean, const at::Tensor& rstd);
std::vector<at::Tensor> addattn_fwd(const at::Tensor& q, const at::Tensor& k,
                                    const at::Tensor& bh, const at::Tensor& wa,
                                    const at::Tensor& ba, long tril_k,
                                    long triu_k);
std::vector<at::Tensor> addattn_bwd(const at::Tensor& q, const at::Tensor& k,
                                    const at::Tensor& bh, const at::Tensor& wa,
                                    const at::Tensor& attn,
                                    const at::Tensor& dattn,
                                    const at::Tensor& amax);
std::vector<at::Tensor> lstm_fwd(const at::Tensor& pre, const at::Tensor& whh,
                                 at::Tensor y, long dir, bool training);
at::Tensor lstm_bwd(const at::Tensor& dy, const at::Tensor& y,
                    const at::Tensor& cstash, const at::Tensor& gstash,
                    const at::Tensor& whh, long dirs, long dir);

at::Tensor adam_pack(std::vector<at::Tensor> params,
                     std::vector<at::Tensor> grads,
                     std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                     std::vector<at::Tensor> masters, bool has_master);
at::Tensor sum_batch(const at::Tensor& in);
at::Tensor channel_sum(const at::Tensor& in);
std::vector<at::Tensor> droppath_dropout_add(const at::Tensor& x,
                                             const at::Tensor& y,
                                             double path_p, double drop_p,
                                             long base);
at::Tensor droppath_dropout_scale(const at::Tensor& dz,
                                  const at::Tensor& slot, double path_p,
                                  double drop_p, long base);
at::Tensor row_scale_add(const at::Tensor& x, const at::Tensor& y,
                         const c10::optional<at::Tensor>& mask, double scale);
at::Tensor row_scale(const at::Tensor& y,
                     const c10::optional<at::Tensor>& mask, double scale);

void adam_step_packed(const at::Tensor& meta, const at::Tensor& sample,
                      bool has_master, double lr, double beta1, double beta2,
                      double eps, double wd, double bc1, double bc2,
                      bool adamw);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("pw_conv_fwd", &pw_conv_fwd, "pointwise conv forward (MFMA GEMM)");
  m.def("pw_conv_multi_fwd", &pw_conv_multi_fwd,
        "pointwise conv over a virtual channel-concat of inputs");
  m.def("pw_conv_multi_dx", &pw_conv_multi_dx,
        "input-gradient written into per-input contiguous tensors");
  m.def("pw_conv_bwd", &pw_conv_bwd, "pointwise conv backward");
  m.def("conv1d_fwd", &conv1d_fwd, "direct conv1d forward");
  m.def("conv1d_bwd", &conv1d_bwd, "direct conv1d backward");
  m.def("conv_transpose1d_fwd", &conv_transpose1d_fwd,
        "transposed conv1d forward");
  m.def("conv_transpose1d_bwd", &conv_transpose1d_bwd,
        "transposed conv1d backward");
  m.def("bn_act_fwd", &bn_act_fwd, "fused batchnorm+act forward");
  m.def("bn_act_bwd", &bn_act_bwd, "fused batchnorm+act backward");
  m.def("bn_sums_only", &bn_sums_only, "local BN (sum, sumsq) to (C,2)");
  m.def("pw_conv_fwd_stats", &pw_conv_fwd_stats,
        "pointwise conv forward + BN stats partials (fusion step 1)");
  m.def("conv1d_fwd_stats", &conv1d_fwd_stats,
        "conv1d forward + BN stats partials (fusion step 1)");
  m.def("bn_act_fwd_with_part", &bn_act_fwd_with_part,
        "BN+act forward from producer-collected partials");
  m.def("bn_act_cat_fwd", &bn_act_cat_fwd,
        "BN+act over a virtual channel-concat (fwd)");
  m.def("bn_act_cat_bwd", &bn_act_cat_bwd,
        "BN+act concat backward (per-input contiguous dx)");
  m.def("bn_part_to_sums", &bn_part_to_sums,
        "(C,nsplit,2) partial slab -> (C,2) sums");
  m.def("bn_finalize_only", &bn_finalize_only,
        "mean/invstd + running update from (C,2) sums");
  m.def("pw_conv_pre_fwd", &pw_conv_pre_fwd,
        "pointwise conv with BN/act transform applied in staging");
  m.def("pw_conv_dx", &pw_conv_dx, "pointwise input-gradient only");
  m.def("bn_bwd_dx_eval", &bn_bwd_dx_eval,
        "eval-mode BN/act dx only (no dgamma/dbeta)");
  m.def("pw_dw_pre", &pw_dw_pre,
        "split-K MFMA weight grad with transform-staged x");
  m.def("bn_act_fwd_from_sums", &bn_act_fwd_from_sums,
        "BN+act forward from externally reduced sums (SyncBN)");
  m.def("bn_bwd_sums_only", &bn_bwd_sums_only,
        "local BN backward (dbeta, dgamma) sums to (C,2)");
  m.def("bn_bwd_dx_from_sums", &bn_bwd_dx_from_sums,
        "BN backward dx from externally reduced sums (SyncBN)");
  m.def("avgmax_pool_fwd", &avgmax_pool_fwd, "fused avg+max pool forward");
  m.def("max_pool1d_fwd", &max_pool1d_fwd, "max pool (stride=k) forward");
  m.def("max_pool1d_bwd", &max_pool1d_bwd, "max pool backward");
  m.def("gap_fwd", &gap_fwd, "global average pool forward");
  m.def("gap_bwd", &gap_bwd, "global average pool backward");
  m.def("loss_sum_fwd", &loss_sum_fwd, "fused BCE/CE loss forward");
  m.def("loss_sum_bwd", &loss_sum_bwd, "fused BCE/CE loss backward");
  m.def("avgmax_pool_bwd", &avgmax_pool_bwd, "fused avg+max pool backward");
  m.def("interp_linear_fwd", &interp_linear_fwd, "linear interp forward");
  m.def("interp_linear_bwd", &interp_linear_bwd, "linear interp backward");
  m.def("upsample2x_fwd", &upsample2x_fwd, "nearest 2x upsample forward");
  m.def("upsample2x_bwd", &upsample2x_bwd, "nearest 2x upsample backward");
  m.def("pooled_attn_fwd", &pooled_attn_fwd,
        "fused pooled-KV attention forward (inference)");
  m.def("pooled_attn_train_fwd", &pooled_attn_train_fwd,
        "fused pooled-KV attention training forward (stats + packed mask)");
  m.def("pooled_attn_bwd", &pooled_attn_bwd,
        "fused pooled-KV attention backward (dq, dk, dv)");
  m.def("ln_fwd", &ln_fwd, "LayerNorm forward (channel-last rows)");
  m.def("ln_bwd", &ln_bwd, "LayerNorm backward");
  m.def("addattn_fwd", &addattn_fwd,
        "fused additive attention scores+softmax (EQT K10)");
  m.def("addattn_bwd", &addattn_bwd, "additive attention backward");
  m.def("lstm_fwd", &lstm_fwd, "LSTM recurrence forward (persistent)");
  m.def("lstm_bwd", &lstm_bwd, "LSTM recurrence backward (BPTT)");
  m.def("adam_pack", &adam_pack, "pack fused-adam chunk metadata");
  m.def("sum_batch", &sum_batch, "batch-axis sum to fp32");
  m.def("channel_sum", &channel_sum, "per-channel sum to fp32");
  m.def("row_scale_add", &row_scale_add, "z = x + mask[n]*scale*y");
  m.def("droppath_dropout_add", &droppath_dropout_add,
        "fused residual + DropPath + Dropout (replay-safe seed slot)");
  m.def("droppath_dropout_scale", &droppath_dropout_scale,
        "backward mask-scale for droppath_dropout_add");
  m.def("row_scale", &row_scale, "z = mask[n]*scale*y");
  m.def("adam_step_packed", &adam_step_packed, "fused adam step");
}
