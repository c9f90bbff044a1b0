// fedkit._C — pybind bindings for the CDNA4 (gfx950) kernel library.
// Build: __graft_entry__.build() -> setup.py build_ext --inplace with
// PYTORCH_ROCM_ARCH=gfx950 (in-tree fedkit/_C.so; no JIT cache).

#include <torch/extension.h>
#include <vector>

at::Tensor fedkit_elu_fwd(const at::Tensor& x);
std::vector<at::Tensor> fedkit_max_pool2d_fwd(const at::Tensor& x, long k);
at::Tensor fedkit_max_pool2d_bwd(const at::Tensor& gy, const at::Tensor& idx,
                                 long k, long H, long W);
at::Tensor fedkit_avg_pool2d_fwd(const at::Tensor& x, long k);
at::Tensor fedkit_avg_pool2d_bwd(const at::Tensor& gy, long k, long H,
                                 long W);
at::Tensor fedkit_elu_bwd(const at::Tensor& gy, const at::Tensor& y);

void fedkit_pack_params(std::vector<at::Tensor> tensors, at::Tensor flat);
void fedkit_unpack_params(at::Tensor flat, std::vector<at::Tensor> tensors);
void fedkit_add_flat_params(std::vector<at::Tensor> tensors, at::Tensor flat,
                            double alpha);
at::Tensor fedkit_multi_dot(std::vector<at::Tensor> vecs, const at::Tensor& x);
at::Tensor fedkit_lincomb(const at::Tensor& g, double cg,
                          std::vector<at::Tensor> vecs,
                          std::vector<double> coeffs);
void fedkit_cast_f32_to_bf16(std::vector<at::Tensor> srcs,
                             std::vector<at::Tensor> dsts);
void fedkit_cast_bf16_to_f32(std::vector<at::Tensor> srcs,
                             std::vector<at::Tensor> dsts);

std::vector<at::Tensor> fedkit_cross_entropy_fwd(const at::Tensor& logits,
                                                 const at::Tensor& labels);
at::Tensor fedkit_cross_entropy_bwd(const at::Tensor& logits,
                                    const at::Tensor& labels,
                                    const at::Tensor& lse,
                                    const at::Tensor& gloss);

std::vector<at::Tensor> fedkit_bn_fwd(const at::Tensor& x,
                                      const at::Tensor& gamma,
                                      const at::Tensor& beta,
                                      at::Tensor running_mean,
                                      at::Tensor running_var, bool training,
                                      double momentum, double eps,
                                      c10::optional<at::Tensor> residual,
                                      bool elu,
                                      c10::optional<at::Tensor> conv_part,
                                      long pad_out, long res_pad);
std::vector<at::Tensor> fedkit_bn_bwd(const at::Tensor& gy, const at::Tensor& x,
                                      const at::Tensor& gamma,
                                      const at::Tensor& save_mean,
                                      const at::Tensor& save_invstd,
                                      c10::optional<at::Tensor> elu_y,
                                      bool want_g, long pad_in);

at::Tensor fedkit_conv2d_fwd(const at::Tensor& x, const at::Tensor& w,
                             long stride, long padding, long dil, long ktrue);
at::Tensor fedkit_conv2d_pad_input(const at::Tensor& x, long padding);
at::Tensor fedkit_dilate_pad(const at::Tensor& x, long pt, long pb, long pl,
                             long pr, long str);
at::Tensor fedkit_conv2d_fwd_prepadded(const at::Tensor& xp,
                                       const at::Tensor& w, long stride,
                                       long dil, long ktrue);
std::vector<at::Tensor> fedkit_conv2d_fwd_prepadded_bnstats(
    const at::Tensor& xp, const at::Tensor& w, long stride);
at::Tensor fedkit_conv2d_bwd_weight_prepadded(const at::Tensor& gy,
                                              const at::Tensor& xp,
                                              long stride, long R, long S,
                                              long dil);
at::Tensor fedkit_conv2d_bwd_data(const at::Tensor& gy, const at::Tensor& w,
                                  long stride, long padding, long H, long W,
                                  long dil, long ctrue);
at::Tensor fedkit_conv2d_bwd_weight(const at::Tensor& gy, const at::Tensor& x,
                                    long stride, long padding, long R, long S,
                                    long dil);

void fedkit_adam_step(std::vector<at::Tensor> params,
                      std::vector<at::Tensor> grads,
                      std::vector<at::Tensor> exp_avg,
                      std::vector<at::Tensor> exp_avg_sq, double lr,
                      double beta1, double beta2, double eps, long step,
                      double weight_decay);
at::Tensor fedkit_welford_update(const at::Tensor& g, at::Tensor avg,
                                 at::Tensor avg_sq, double inv_n);
at::Tensor fedkit_linear_fwd(const at::Tensor& x, const at::Tensor& w,
                             const c10::optional<at::Tensor>& bias);
at::Tensor fedkit_linear_bwd_data(const at::Tensor& gy, const at::Tensor& w);
std::vector<at::Tensor> fedkit_linear_bwd_weight(const at::Tensor& gy,
                                                 const at::Tensor& x,
                                                 bool want_bias);

at::Tensor fedkit_conv2d_dilated_bank(const at::Tensor& x,
                                      const at::Tensor& w2d,
                                      std::vector<long> dils,
                                      std::vector<long> pads, long stride,
                                      long R, long ktrue);

at::Tensor fedkit_vae_elbo_fwd(const at::Tensor& recon, const at::Tensor& x,
                               const at::Tensor& mu, const at::Tensor& logvar);
std::vector<at::Tensor> fedkit_vae_elbo_bwd(const at::Tensor& recon,
                                            const at::Tensor& x,
                                            const at::Tensor& mu,
                                            const at::Tensor& logvar,
                                            const at::Tensor& gloss);
std::vector<at::Tensor> fedkit_vaecl_terms_fwd(
    const at::Tensor& x, const at::Tensor& mu_th, const at::Tensor& s_th,
    const at::Tensor& mu_q, const at::Tensor& s_q, const at::Tensor& mu_p,
    const at::Tensor& s_p, long B);
std::vector<at::Tensor> fedkit_vaecl_terms_bwd(
    const at::Tensor& x, const at::Tensor& mu_th, const at::Tensor& s_th,
    const at::Tensor& mu_q, const at::Tensor& s_q, const at::Tensor& mu_p,
    const at::Tensor& s_p, const at::Tensor& gR1, const at::Tensor& gR3,
    long B);
std::vector<at::Tensor> fedkit_info_nce_fwd(const at::Tensor& Z,
                                            const at::Tensor& Zhat);
std::vector<at::Tensor> fedkit_info_nce_bwd(const at::Tensor& Z,
                                            const at::Tensor& Zhat,
                                            const at::Tensor& dzz,
                                            const at::Tensor& tu,
                                            const at::Tensor& norms);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "fedkit hand-written CDNA4 (gfx950 / MI355X) kernels";
  m.def("elu_fwd", &fedkit_elu_fwd, "ELU forward (vectorized)");
  m.def("max_pool2d_fwd", &fedkit_max_pool2d_fwd,
        "NHWC non-overlapping max pool: returns (y, argmax_idx)");
  m.def("max_pool2d_bwd", &fedkit_max_pool2d_bwd, "max pool backward");
  m.def("avg_pool2d_fwd", &fedkit_avg_pool2d_fwd,
        "NHWC non-overlapping average pool");
  m.def("avg_pool2d_bwd", &fedkit_avg_pool2d_bwd, "avg pool backward");
  m.def("elu_bwd", &fedkit_elu_bwd, "ELU backward from saved output");
  m.def("pack_params", &fedkit_pack_params, "multi-tensor -> flat fp32");
  m.def("unpack_params", &fedkit_unpack_params, "flat fp32 -> multi-tensor");
  m.def("add_flat_params", &fedkit_add_flat_params,
        "t += alpha * flat slice (multi-tensor axpy)");
  m.def("multi_dot", &fedkit_multi_dot,
        "x . vecs[i] for up to 24 vectors in one pass -> device fp32 [n]");
  m.def("lincomb", &fedkit_lincomb,
        "cg*g + sum c[i]*vecs[i] in one pass (L-BFGS direction build)");
  m.def("cast_f32_to_bf16", &fedkit_cast_f32_to_bf16,
        "batched fp32->bf16 tensor casts (one kernel for <=48 tensors)");
  m.def("cast_bf16_to_f32", &fedkit_cast_bf16_to_f32,
        "batched bf16->fp32 tensor casts");
  m.def("cross_entropy_fwd", &fedkit_cross_entropy_fwd,
        "fused log-softmax + NLL (mean): returns (loss, lse)");
  m.def("cross_entropy_bwd", &fedkit_cross_entropy_bwd, "CE backward");
  m.def("bn_fwd", &fedkit_bn_fwd,
        "NHWC BatchNorm fwd (optional fused residual add + ELU epilogue): "
        "returns (y, save_mean, save_invstd)",
        py::arg("x"), py::arg("gamma"), py::arg("beta"),
        py::arg("running_mean"), py::arg("running_var"), py::arg("training"),
        py::arg("momentum"), py::arg("eps"),
        py::arg("residual") = c10::nullopt, py::arg("elu") = false,
        py::arg("conv_part") = c10::nullopt, py::arg("pad_out") = 0,
        py::arg("res_pad") = 0);
  m.def("bn_bwd", &fedkit_bn_bwd,
        "NHWC BatchNorm bwd (optionally fused with ELU backward from the "
        "saved output): returns (gx, gw, gb[, g])",
        py::arg("gy"), py::arg("x"), py::arg("gamma"), py::arg("save_mean"),
        py::arg("save_invstd"), py::arg("elu_y") = c10::nullopt,
        py::arg("want_g") = false, py::arg("pad_in") = 0);
  m.def("conv2d_fwd", &fedkit_conv2d_fwd,
        "NHWC implicit-GEMM conv fwd on MFMA (square filter, stride 1/2, "
        "optional dilation; ktrue marks channel-padded Kout)",
        py::arg("x"), py::arg("w"), py::arg("stride"), py::arg("padding"),
        py::arg("dil") = 1, py::arg("ktrue") = -1);
  m.def("conv2d_pad_input", &fedkit_conv2d_pad_input, "zero-pad NHWC input");
  m.def("conv2d_fwd_prepadded_bnstats", &fedkit_conv2d_fwd_prepadded_bnstats,
        "conv fwd that also emits BatchNorm stage-1 partials from the "
        "epilogue registers: returns (y, part)");
  m.def("dilate_pad", &fedkit_dilate_pad,
        "NHWC zero-insert dilation + border pad (transposed-conv input)");
  m.def("conv2d_fwd_prepadded", &fedkit_conv2d_fwd_prepadded,
        "conv fwd on a pre-padded input",
        py::arg("xp"), py::arg("w"), py::arg("stride"),
        py::arg("dil") = 1, py::arg("ktrue") = -1);
  m.def("conv2d_bwd_weight_prepadded", &fedkit_conv2d_bwd_weight_prepadded,
        "conv bwd-weight from the saved padded input",
        py::arg("gy"), py::arg("xp"), py::arg("stride"), py::arg("R"),
        py::arg("S"), py::arg("dil") = 1);
  m.def("conv2d_bwd_data", &fedkit_conv2d_bwd_data, "conv bwd-data",
        py::arg("gy"), py::arg("w"), py::arg("stride"), py::arg("padding"),
        py::arg("H"), py::arg("W"), py::arg("dil") = 1,
        py::arg("ctrue") = -1);
  m.def("conv2d_bwd_weight", &fedkit_conv2d_bwd_weight, "conv bwd-weight",
        py::arg("gy"), py::arg("x"), py::arg("stride"), py::arg("padding"),
        py::arg("R"), py::arg("S"), py::arg("dil") = 1);
  m.def("adam_step", &fedkit_adam_step,
        "fused Adam: one kernel updates every parameter (fp32 master)");
  m.def("welford_update", &fedkit_welford_update,
        "fused LBFGS Welford grad-stats update; returns sum(avg_sq)");
  m.def("linear_fwd", &fedkit_linear_fwd,
        "fused linear fwd (x [M,K], w [N,K], bias) -> [M,N]",
        py::arg("x"), py::arg("w"), py::arg("bias") = c10::nullopt);
  m.def("linear_bwd_data", &fedkit_linear_bwd_data, "linear bwd-data");
  m.def("linear_bwd_weight", &fedkit_linear_bwd_weight,
        "linear bwd-weight (+bias): fp32 grads",
        py::arg("gy"), py::arg("x"), py::arg("want_bias") = true);
  m.def("conv2d_dilated_bank", &fedkit_conv2d_dilated_bank,
        "fused multi-dilation conv bank: n taps, one input, one launch "
        "(block-diagonal combined weight)");
  m.def("vae_elbo_fwd", &fedkit_vae_elbo_fwd,
        "MSE(sum) + analytic KLD in one reduction pass");
  m.def("vae_elbo_bwd", &fedkit_vae_elbo_bwd, "VAE ELBO backward");
  m.def("vaecl_terms_fwd", &fedkit_vaecl_terms_fwd,
        "VAE-CL per-(cluster,sample) cost1/cost3 reductions: (R1, R3)");
  m.def("vaecl_terms_bwd", &fedkit_vaecl_terms_bwd,
        "VAE-CL terms backward (elementwise)");
  m.def("info_nce_fwd", &fedkit_info_nce_fwd,
        "fused InfoNCE: returns (loss, zz, softmax, norms)");
  m.def("info_nce_bwd", &fedkit_info_nce_bwd, "InfoNCE backward");
}
