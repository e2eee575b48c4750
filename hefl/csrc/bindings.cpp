// hefl._C — torch extension binding the gfx950 HIP kernels:
// CKKS NTT/pointwise (ntt.hip) + CNN training ops (cnn.hip).
#include <torch/extension.h>

#include <vector>

// ntt.hip
void ntt_batch(torch::Tensor x, torch::Tensor w, torch::Tensor wsh, int64_t q);
void intt_batch(torch::Tensor x, torch::Tensor winv, torch::Tensor winvsh,
                int64_t q, int64_t ninv, int64_t ninvsh);
torch::Tensor modmul(torch::Tensor a, torch::Tensor b, int64_t q);
torch::Tensor modmul_scalar(torch::Tensor a, int64_t s, int64_t q);
torch::Tensor modadd(torch::Tensor a, torch::Tensor b, int64_t q);
torch::Tensor modsub(torch::Tensor a, torch::Tensor b, int64_t q);
void modreduce_(torch::Tensor x, torch::Tensor qs);
torch::Tensor cbd21(torch::Tensor bits);
void ntt_limbs(torch::Tensor x, torch::Tensor w, torch::Tensor wsh,
               torch::Tensor qs, int64_t L);
void intt_limbs(torch::Tensor x, torch::Tensor winv, torch::Tensor winvsh,
                torch::Tensor qs, torch::Tensor ninv, torch::Tensor ninvsh,
                int64_t L);
torch::Tensor modmul_limbs(torch::Tensor a, torch::Tensor b, torch::Tensor qs,
                           torch::Tensor ratios, int64_t L, int64_t n);
torch::Tensor modmul_scalar_limbs(torch::Tensor a, torch::Tensor scalars,
                                  torch::Tensor shoups, torch::Tensor qs,
                                  int64_t L, int64_t n);
torch::Tensor modadd_limbs(torch::Tensor a, torch::Tensor b, torch::Tensor qs,
                           int64_t L, int64_t n);
torch::Tensor modadd3_limbs(torch::Tensor a, torch::Tensor b, torch::Tensor c,
                            torch::Tensor qs, int64_t L, int64_t n);
torch::Tensor modsub_limbs(torch::Tensor a, torch::Tensor b, torch::Tensor qs,
                           int64_t L, int64_t aL, int64_t n);
torch::Tensor bcast_center_mod(torch::Tensor x, int64_t qc, torch::Tensor qs,
                               torch::Tensor ratios, int64_t L);
std::vector<torch::Tensor> ct_mul(torch::Tensor a, torch::Tensor b,
                                  torch::Tensor qs, torch::Tensor ratios,
                                  int64_t L, int64_t n);
std::vector<torch::Tensor> ks_inner(torch::Tensor dig, torch::Tensor rlk,
                                    torch::Tensor qs, torch::Tensor ratios,
                                    int64_t D, int64_t Lp, int64_t n);

// fft.hip
torch::Tensor fft_encode(torch::Tensor vals, torch::Tensor tw_enc,
                         double scale);
torch::Tensor fft_decode(torch::Tensor coeffs, torch::Tensor tw_dec,
                         double scale, int64_t k);

// cnn.hip
torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                         int64_t stride, bool relu, int64_t pad);
torch::Tensor conv2d_dgrad(torch::Tensor dy, torch::Tensor w, int64_t stride,
                           int64_t H, int64_t W, int64_t pad);
torch::Tensor conv2d_wgrad(torch::Tensor dy, torch::Tensor x, int64_t stride,
                           int64_t R, int64_t S, int64_t pad, int64_t gkey);
torch::Tensor linear_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                         bool relu);
torch::Tensor linear_dgrad(torch::Tensor dy, torch::Tensor w);
torch::Tensor linear_wgrad(torch::Tensor dy, torch::Tensor x);
std::vector<torch::Tensor> maxpool2x2_fwd(torch::Tensor x);
torch::Tensor synth_batch(torch::Tensor templates, torch::Tensor labels,
                          int64_t seed, double zoom, double shear,
                          int64_t flip);
torch::Tensor synth_batch_g(torch::Tensor templates, torch::Tensor labels,
                            torch::Tensor seed_buf, int64_t salt, double zoom,
                            double shear, int64_t flip);
torch::Tensor maxpool2x2_bwd(torch::Tensor dy, torch::Tensor idx, int64_t H,
                             int64_t W);
std::vector<torch::Tensor> softmax_xent_fwd(torch::Tensor logits,
                                            torch::Tensor labels,
                                            torch::Tensor acc_loss,
                                            torch::Tensor acc_correct);
torch::Tensor softmax_xent_bwd(torch::Tensor probs, torch::Tensor labels,
                               torch::Tensor dloss, bool out_bf16);
void fused_adam(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, double lr, double b1, double b2, double eps,
                double bc1, double bc2);
void adam_prep(torch::Tensor step, torch::Tensor sched, torch::Tensor hyper,
               double b1, double b2);
torch::Tensor relu_bwd(torch::Tensor dy, torch::Tensor y);
std::vector<torch::Tensor> bn_fwd(torch::Tensor x, torch::Tensor gamma,
                                  torch::Tensor beta, torch::Tensor rmean,
                                  torch::Tensor rvar, double eps,
                                  double momentum, bool relu,
                                  torch::Tensor sums);
torch::Tensor bn_apply(torch::Tensor x, torch::Tensor mean,
                       torch::Tensor invstd, torch::Tensor gamma,
                       torch::Tensor beta, bool relu);
std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor mean, torch::Tensor invstd,
                                  torch::Tensor gamma, bool train,
                                  torch::Tensor relu_y, torch::Tensor fwd_sums);
std::vector<torch::Tensor> maxpool_fwd(torch::Tensor x, int64_t k, int64_t s,
                                       int64_t p);
torch::Tensor maxpool_bwd(torch::Tensor dy, torch::Tensor idx, int64_t H,
                          int64_t W, int64_t k, int64_t s, int64_t p);
torch::Tensor avgpool_global_fwd(torch::Tensor x);
torch::Tensor avgpool_global_bwd(torch::Tensor dy, int64_t H, int64_t W);
torch::Tensor add_relu(torch::Tensor a, torch::Tensor b);
torch::Tensor bias_grad(torch::Tensor dy, int64_t gkey);
std::vector<torch::Tensor> relu_bias_bwd(torch::Tensor dy, torch::Tensor y,
                                         int64_t gkey);
std::vector<torch::Tensor> pool_relu_bias_bwd(torch::Tensor dy,
                                              torch::Tensor idx,
                                              torch::Tensor p, int64_t H,
                                              int64_t W, int64_t gkey);
void adam_prep_epoch(torch::Tensor step, torch::Tensor sched,
                     torch::Tensor hyper, double b1, double b2, int64_t S);
torch::Tensor pad_channels(torch::Tensor x, int64_t C8);
std::vector<torch::Tensor> dense_head2_fwd(torch::Tensor x,
                                           torch::Tensor w1, torch::Tensor b1,
                                           torch::Tensor w2, torch::Tensor b2);
std::vector<torch::Tensor> dense_head2_bwd(torch::Tensor dlogits,
                                           torch::Tensor x, torch::Tensor h1,
                                           torch::Tensor w1, torch::Tensor w2);
void pack_mt(torch::Tensor meta, torch::Tensor ptrs, torch::Tensor sizes,
             torch::Tensor offs, int64_t nchunks, torch::Tensor flat);
void unpack_mt(torch::Tensor flat, torch::Tensor meta, torch::Tensor ptrs,
               torch::Tensor shptrs, torch::Tensor sizes, torch::Tensor offs,
               int64_t nchunks);
void fused_adam_mt(torch::Tensor meta, torch::Tensor ptrs, torch::Tensor sizes,
                   int64_t nchunks, torch::Tensor sched, double b1, double b2,
                   double eps, int64_t zero_g, int64_t sched_off);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "hefl gfx950 (MI355X/CDNA4) HIP kernels";
    // HE / CKKS
    m.def("ntt_batch", &ntt_batch, "in-place forward negacyclic NTT [rows, n]");
    m.def("intt_batch", &intt_batch, "in-place inverse negacyclic NTT");
    m.def("modmul", &modmul, "pointwise Barrett modmul");
    m.def("modmul_scalar", &modmul_scalar, "pointwise Shoup scalar modmul");
    m.def("modadd", &modadd);
    m.def("modsub", &modsub);
    m.def("modreduce_", &modreduce_, "in-place per-limb reduction after lazy sum");
    m.def("cbd21", &cbd21, "centered binomial eta=21 from 64-bit draws");
    m.def("ntt_limbs", &ntt_limbs, "fused multi-limb forward NTT [R, L, n]");
    m.def("intt_limbs", &intt_limbs, "fused multi-limb inverse NTT");
    m.def("modmul_limbs", &modmul_limbs);
    m.def("modmul_scalar_limbs", &modmul_scalar_limbs);
    m.def("modadd_limbs", &modadd_limbs, "per-limb modular add over [.., L, n]");
    m.def("modadd3_limbs", &modadd3_limbs, "fused 3-way per-limb modular add");
    m.def("modsub_limbs", &modsub_limbs, "per-limb modular sub over [.., L, n]");
    m.def("bcast_center_mod", &bcast_center_mod,
          "[.., n] -> [.., L, n] center + per-limb Barrett reduce");
    m.def("ct_mul", &ct_mul, "fused ct x ct tensor product (d0, d1, d2)");
    m.def("ks_inner", &ks_inner, "fused key-switch digit inner product");
    m.def("fft_encode", &fft_encode,
          "CKKS special-FFT encode: f64 slots -> int64 coeffs");
    m.def("fft_decode", &fft_decode,
          "CKKS special-FFT decode: int64 coeffs -> f32 slots");
    // CNN
    m.def("conv2d_fwd", &conv2d_fwd);
    m.def("conv2d_dgrad", &conv2d_dgrad);
    m.def("conv2d_wgrad", &conv2d_wgrad, py::arg("dy"), py::arg("x"),
          py::arg("stride"), py::arg("R"), py::arg("S"), py::arg("pad"),
          py::arg("gkey") = 0);
    m.def("linear_fwd", &linear_fwd);
    m.def("linear_dgrad", &linear_dgrad);
    m.def("linear_wgrad", &linear_wgrad);
    m.def("maxpool2x2_fwd", &maxpool2x2_fwd);
    m.def("synth_batch", &synth_batch, py::arg("templates"), py::arg("labels"),
          py::arg("seed"), py::arg("zoom") = 0.0, py::arg("shear") = 0.0,
          py::arg("flip") = 0);
    m.def("synth_batch_g", &synth_batch_g, py::arg("templates"),
          py::arg("labels"), py::arg("seed_buf"), py::arg("salt"),
          py::arg("zoom") = 0.0, py::arg("shear") = 0.0, py::arg("flip") = 0);
    m.def("maxpool2x2_bwd", &maxpool2x2_bwd);
    m.def("softmax_xent_fwd", &softmax_xent_fwd);
    m.def("softmax_xent_bwd", &softmax_xent_bwd);
    m.def("fused_adam", &fused_adam);
    m.def("adam_prep", &adam_prep);
    m.def("relu_bwd", &relu_bwd);
    m.def("bn_fwd", &bn_fwd);
    m.def("bn_apply", &bn_apply);
    m.def("bn_bwd", &bn_bwd);
    m.def("maxpool_fwd", &maxpool_fwd);
    m.def("maxpool_bwd", &maxpool_bwd);
    m.def("avgpool_global_fwd", &avgpool_global_fwd);
    m.def("avgpool_global_bwd", &avgpool_global_bwd);
    m.def("add_relu", &add_relu);
    m.def("bias_grad", &bias_grad, py::arg("dy"), py::arg("gkey") = 0);
    m.def("relu_bias_bwd", &relu_bias_bwd, py::arg("dy"), py::arg("y"),
          py::arg("gkey") = 0);
    m.def("pool_relu_bias_bwd", &pool_relu_bias_bwd, py::arg("dy"),
          py::arg("idx"), py::arg("p"), py::arg("H"), py::arg("W"),
          py::arg("gkey") = 0);
    m.def("fused_adam_mt", &fused_adam_mt);
    m.def("pad_channels", &pad_channels,
          "zero-pad NHWC channel dim to C8 (stem -> glds MFMA path)");
    m.def("dense_head2_fwd", &dense_head2_fwd);
    m.def("dense_head2_bwd", &dense_head2_bwd,
          "single-launch backward of the 2-layer dense head (M <= 32)");
    m.def("pack_mt", &pack_mt);
    m.def("unpack_mt", &unpack_mt);
    m.def("adam_prep_epoch", &adam_prep_epoch);
}
