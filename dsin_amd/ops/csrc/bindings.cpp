// Python bindings for the dsin_amd HIP/CDNA4 kernels.
#include <torch/extension.h>

namespace dsin {
std::tuple<torch::Tensor, torch::Tensor> quantize_fwd(torch::Tensor x,
                                                      torch::Tensor centers,
                                                      double sigma);
std::tuple<torch::Tensor, torch::Tensor> quantize_bwd(torch::Tensor g,
                                                      torch::Tensor x,
                                                      torch::Tensor centers,
                                                      double sigma);
torch::Tensor bitcost_ce_fwd(torch::Tensor logits, torch::Tensor symbols);
torch::Tensor bitcost_ce_bwd(torch::Tensor g, torch::Tensor logits,
                             torch::Tensor symbols);
std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> ncc_search(
    torch::Tensor x_dec, torch::Tensor y_dec, torch::Tensor y_orig, int64_t ph,
    int64_t pw, bool use_mask);
torch::Tensor mfma_selftest(torch::Tensor A, torch::Tensor B);
torch::Tensor mfma32_selftest(torch::Tensor A, torch::Tensor B);
std::tuple<torch::Tensor, torch::Tensor> conv_tables(
    int64_t M, int64_t K, int64_t WO, int64_t stride, int64_t dil, int64_t Wp,
    int64_t HpWp, int64_t kh, int64_t kw, torch::Device device);
std::tuple<torch::Tensor, torch::Tensor> conv_tables_v2(
    int64_t M, int64_t K, int64_t WO, int64_t stride, int64_t dil, int64_t kh,
    int64_t kw, torch::Device device);
torch::Tensor conv_fwd(torch::Tensor xbuf, torch::Tensor wmat,
                       c10::optional<torch::Tensor> bias, torch::Tensor mbase,
                       torch::Tensor koff, int64_t N, int64_t K, int64_t HO,
                       int64_t WO, int64_t act, int64_t stride, int64_t direct,
                       int64_t vpad, int64_t vm, int64_t vpt, int64_t vpl,
                       int64_t vsv, c10::optional<torch::Tensor> out_opt,
                       int64_t oh0, int64_t ow0, int64_t ostep, int64_t wof);
torch::Tensor panel_gather(torch::Tensor src, torch::Tensor ktab);
torch::Tensor conv_wrw(torch::Tensor xbuf, torch::Tensor dy,
                       torch::Tensor mbase, torch::Tensor koff, int64_t N,
                       int64_t K, int64_t WO, bool mcontig, int64_t vm,
                       int64_t st, int64_t vpt, int64_t vpl, int64_t vsv);
torch::Tensor act_bwd(torch::Tensor dy, torch::Tensor y, int64_t act);
torch::Tensor wmat_make(torch::Tensor w1, int64_t khw, int64_t mode);
void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, torch::Tensor lr, torch::Tensor step,
               c10::optional<torch::Tensor> wd, double b1, double b2,
               double eps);
torch::Tensor pad_stuff(torch::Tensor x, int64_t pt, int64_t pb, int64_t pl,
                        int64_t pr, int64_t stride, bool fp8);
std::vector<torch::Tensor> bn_fwd(torch::Tensor y, torch::Tensor gamma,
                                  torch::Tensor beta, torch::Tensor rmean,
                                  torch::Tensor rvar, double momentum,
                                  double eps, bool training, int64_t act,
                                  c10::optional<torch::Tensor> residual);
std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor y,
                                  torch::Tensor out, torch::Tensor mean,
                                  torch::Tensor rstd, torch::Tensor gamma,
                                  bool training, int64_t act);
std::vector<torch::Tensor> heatmap_mask_fwd(torch::Tensor b);
torch::Tensor heatmap_mask_bwd(torch::Tensor b, torch::Tensor gz,
                               c10::optional<torch::Tensor> gh3);
torch::Tensor l1_part(torch::Tensor x, torch::Tensor y);
std::vector<torch::Tensor> l1_bwd(torch::Tensor x, torch::Tensor y,
                                  torch::Tensor g, bool need_gx,
                                  bool need_gy);
torch::Tensor hterms_part(torch::Tensor bc, torch::Tensor heat);
std::vector<torch::Tensor> hterms_bwd(torch::Tensor bc, torch::Tensor heat,
                                      torch::Tensor g2, bool need_gheat);
}  // namespace dsin

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("quantize_fwd", &dsin::quantize_fwd, "fused soft quantizer forward");
  m.def("quantize_bwd", &dsin::quantize_bwd, "fused soft quantizer backward");
  m.def("bitcost_ce_fwd", &dsin::bitcost_ce_fwd, "bitcost cross-entropy fwd");
  m.def("bitcost_ce_bwd", &dsin::bitcost_ce_bwd, "bitcost cross-entropy bwd");
  m.def("ncc_search", &dsin::ncc_search, "streaming NCC side-info search");
  m.def("mfma_selftest", &dsin::mfma_selftest, "MFMA 16x16x32 layout check");
  m.def("mfma32_selftest", &dsin::mfma32_selftest, "MFMA 32x32x16 layout check");
  m.def("conv_tables", &dsin::conv_tables, "gather-conv offset tables");
  m.def("conv_tables_v2", &dsin::conv_tables_v2,
        "packed-coordinate tables for the virtual-pad gather path");
  m.def("conv_fwd", &dsin::conv_fwd, "implicit-GEMM gather conv forward");
  m.def("conv_wrw", &dsin::conv_wrw, "implicit-GEMM conv weight gradient");
  m.def("act_bwd", &dsin::act_bwd, "fused activation gradient (bf16 out)");
  m.def("wmat_make", &dsin::wmat_make, "padded/rotated bf16 conv W panel");
  m.def("panel_gather", &dsin::panel_gather,
        "column-gather of a W panel (phase-decomposed convs)");
  m.def("adam_step", &dsin::adam_step, "fused flat-buffer Adam step");
  m.def("pad_stuff", &dsin::pad_stuff, "fused pad/zero-stuff/cast to bf16");
  m.def("bn_fwd", &dsin::bn_fwd, "fused batch-norm(+act) forward");
  m.def("bn_bwd", &dsin::bn_bwd, "fused batch-norm(+act) backward");
  m.def("heatmap_mask_fwd", &dsin::heatmap_mask_fwd,
        "fused heatmap3D + bottleneck mask");
  m.def("heatmap_mask_bwd", &dsin::heatmap_mask_bwd,
        "heatmap3D + mask backward");
  m.def("l1_part", &dsin::l1_part, "per-image |y-x| partial sums");
  m.def("l1_bwd", &dsin::l1_bwd, "L1-mean backward");
  m.def("hterms_part", &dsin::hterms_part,
        "fused (sum bc, sum bc*heatmap) partials");
  m.def("hterms_bwd", &dsin::hterms_bwd, "rate-term backward");
}
