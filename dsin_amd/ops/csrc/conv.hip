// Host wrappers for the implicit-GEMM conv family (device code in
// conv_kernels.h). See dsin_amd/ops/conv.py for the autograd layer and the
// geometry-to-table mapping.

#include "common.h"
#include "conv_kernels.h"
#define DSIN_CONV_FP8_KERNELS
#include "conv_fp8.h"

namespace dsin {

std::tuple<torch::Tensor, torch::Tensor> conv_tables(
    int64_t M, int64_t K, int64_t WO, int64_t stride, int64_t dil, int64_t Wp,
    int64_t HpWp, int64_t kh, int64_t kw, torch::Device device) {
  auto opts = torch::TensorOptions().device(device).dtype(torch::kInt32);
  auto mbase = torch::empty({M}, opts);
  auto koff = torch::empty({K}, opts);
  int64_t n = std::max(M, K);
  hipLaunchKernelGGL(conv_tables_kernel, grid1d(n, 256), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     mbase.data_ptr<int>(), koff.data_ptr<int>(), (int)M,
                     (int)K, (int)WO, (int)stride, (int)dil, (int)Wp,
                     (int)HpWp, (int)(kh * kw), (int)kw);
  return {mbase, koff};
}

std::tuple<torch::Tensor, torch::Tensor> conv_tables_v2(
    int64_t M, int64_t K, int64_t WO, int64_t stride, int64_t dil, int64_t kh,
    int64_t kw, torch::Device device) {
  auto opts = torch::TensorOptions().device(device).dtype(torch::kInt32);
  auto mpack = torch::empty({M}, opts);
  auto kpack = torch::empty({K}, opts);
  int64_t n = std::max(M, K);
  hipLaunchKernelGGL(conv_tables_v2_kernel, grid1d(n, 256), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), mpack.data_ptr<int>(),
                     kpack.data_ptr<int>(), (int)M, (int)K, (int)WO,
                     (int)stride, (int)dil, (int)(kh * kw), (int)kw);
  return {mpack, kpack};
}

torch::Tensor conv_fwd(torch::Tensor xbuf, torch::Tensor wmat,
                       c10::optional<torch::Tensor> bias,
                       torch::Tensor mbase, torch::Tensor koff, int64_t N,
                       int64_t K, int64_t HO, int64_t WO, int64_t act,
                       int64_t stride, int64_t direct, int64_t vpad,
                       int64_t vm, int64_t vpt, int64_t vpl, int64_t vsv,
                       c10::optional<torch::Tensor> out_opt, int64_t oh0,
                       int64_t ow0, int64_t ostep, int64_t wof) {
  CHECK_CUDA_CONTIG(xbuf);
  CHECK_CUDA_CONTIG(wmat);
  const bool fp8 = xbuf.scalar_type() == torch::kByte;  // raw e4m3 bytes
  TORCH_CHECK(fp8 || xbuf.scalar_type() == torch::kBFloat16,
              "xbuf must be bf16 or e4m3-as-uint8");
  TORCH_CHECK(wmat.scalar_type() == xbuf.scalar_type(), "x/w dtype mismatch");
  const int64_t B = xbuf.size(0);
  const int64_t M = HO * WO;
  if (direct == 2 && !fp8) {
    // 5x5 stride-2 direct LDS-halo kernel (virtual pad)
    const int64_t Ci = xbuf.size(1);
    const int64_t KPd = (K + 63) & ~63;
    TORCH_CHECK(K == Ci * 25 && Ci % 64 == 0 && stride == 2,
                "direct5 conv gate mismatch");
    TORCH_CHECK(wmat.size(1) == KPd + CONV_AP, "direct5 wmat stride mismatch");
    const int Hp = (int)xbuf.size(2), Wp = (int)xbuf.size(3);
    TORCH_CHECK(HO == (Hp + 2 * vpad - 5) / 2 + 1 &&
                    WO == (Wp + 2 * vpad - 5) / 2 + 1,
                "direct5 conv size mismatch");
    auto out = torch::empty({B, N, HO, WO},
                            xbuf.options().dtype(torch::kBFloat16));
    const float* bp = nullptr;
    if (bias.has_value()) {
      CHECK_CUDA_CONTIG(bias.value());
      bp = bias->data_ptr<float>();
    }
    dim3 grid(((HO + 7) / 8) * ((WO + 7) / 8), (N + 63) / 64, B);
    size_t lds = (size_t)19 * 19 * 64 * 2;
    hipLaunchKernelGGL(conv5x5s2_direct_kernel, grid, dim3(256), lds,
                       at::cuda::getCurrentCUDAStream(),
                       (const cvbf16*)xbuf.data_ptr(),
                       (const cvbf16*)wmat.data_ptr(), bp,
                       (cvbf16*)out.data_ptr(), (int)Ci, Hp, Wp, (int)N,
                       (int)HO, (int)WO, (int)KPd, xbuf.stride(0),
                       (long long)N * M, (int)act, (int)vpad);
    return out;
  }
  if (direct && !fp8) {
    // stride-1 3x3, Ci%64==0: direct LDS-halo kernel (9x less gather
    // traffic than the im2col path); caller built wmat in mode-2/3 layout
    const int64_t Ci = xbuf.size(1);
    const int64_t KPd = (K + 63) & ~63;
    TORCH_CHECK(K == Ci * 9 && Ci % 64 == 0 && stride == 1,
                "direct conv gate mismatch");
    TORCH_CHECK(wmat.size(1) == KPd + CONV_AP, "direct wmat stride mismatch");
    auto out = torch::empty({B, N, HO, WO},
                            xbuf.options().dtype(torch::kBFloat16));
    const float* bp = nullptr;
    if (bias.has_value()) {
      CHECK_CUDA_CONTIG(bias.value());
      bp = bias->data_ptr<float>();
    }
    const int Hp = (int)xbuf.size(2), Wp = (int)xbuf.size(3);
    TORCH_CHECK(HO == Hp + 2 * vpad - 2 && WO == Wp + 2 * vpad - 2,
                "direct conv size mismatch");
    dim3 grid(((HO + 7) / 8) * ((WO + 7) / 8), (N + 63) / 64, B);
    size_t lds = (size_t)2 * 10 * 11 * 64 * 2;
    hipLaunchKernelGGL(conv3x3_direct_kernel, grid, dim3(256), lds,
                       at::cuda::getCurrentCUDAStream(),
                       (const cvbf16*)xbuf.data_ptr(),
                       (const cvbf16*)wmat.data_ptr(), bp,
                       (cvbf16*)out.data_ptr(), (int)Ci, Hp, Wp, (int)N,
                       (int)HO, (int)WO, (int)KPd, xbuf.stride(0),
                       (long long)N * M, (int)act, (int)vpad);
    return out;
  }
  const int64_t KP = (K + 63) & ~63;  // 64-chunk padded; wmat zero-padded
  TORCH_CHECK(wmat.size(1) == KP + CONV_AP, "wmat row stride mismatch");
  if (WO < 8) stride = 0;  // scalar staging path for very narrow outputs
  // phase-strided output: write into a caller-provided full-grid tensor
  torch::Tensor out;
  long long o_chan = M, o_img = (long long)N * M;
  int WOf = (int)WO;
  if (out_opt.has_value()) {
    out = out_opt.value();
    CHECK_CUDA_CONTIG(out);
    TORCH_CHECK(out.scalar_type() == torch::kBFloat16 && out.dim() == 4 &&
                    out.size(0) == B && out.size(1) == N,
                "phase out tensor mismatch");
    o_chan = (long long)out.size(2) * out.size(3);
    o_img = (long long)N * o_chan;
    WOf = (int)out.size(3);
  } else {
    TORCH_CHECK(ostep == 1, "strided output needs an out tensor");
    out = torch::empty({B, N, HO, WO},
                       xbuf.options().dtype(torch::kBFloat16));
  }
  const float* bptr = nullptr;
  if (bias.has_value()) {
    CHECK_CUDA_CONTIG(bias.value());
    bptr = bias->data_ptr<float>();
  }
  dim3 grid((M + CONV_TM - 1) / CONV_TM, (N + CONV_TN - 1) / CONV_TN, B);
  if (fp8) {
    TORCH_CHECK(!out_opt.has_value(), "fp8 path has no phase output");
    size_t lds = (size_t)2 * CONV8_TM * (64 + CONV8_AP);
    hipLaunchKernelGGL(conv_fwd_fp8_kernel, grid, dim3(256), lds,
                       at::cuda::getCurrentCUDAStream(),
                       (const f8*)xbuf.data_ptr(), (const f8*)wmat.data_ptr(),
                       bptr, (cvbf16*)out.data_ptr(), mbase.data_ptr<int>(),
                       koff.data_ptr<int>(), (int)M, (int)N, (int)K, (int)KP,
                       xbuf.stride(0), (long long)N * M, (int)act, (int)WO,
                       (int)stride);
    return out;
  }
  // TM=64 is the throughput tile; when its grid would underfill 256 CUs
  // (< ~3 workgroups/CU) halve the pixel tile to double parallelism — the
  // small-M resblock convs (40x120) are latency-bound, not FLOP-bound.
  const int vH = (int)xbuf.size(2), vW = (int)xbuf.size(3);
  using FwdKern = void (*)(const cvbf16*, const cvbf16*, const float*,
                           cvbf16*, const int*, const int*, int, int, int,
                           int, long long, long long, int, int, int, int,
                           int, int, int, int, int, int, int, int,
                           long long);
  auto pick_fwd = [&](bool tm32) -> FwdKern {
    if (!vm) return tm32 ? conv_fwd_kernel<32, 0, 0, 1>
                         : conv_fwd_kernel<64, 0, 0, 1>;
    // compile-time (stride, stuff) specialization for the virtual path
    if (vsv == 2) {
      if (stride == 1)
        return tm32 ? conv_fwd_kernel<32, 1, 1, 2>
                    : conv_fwd_kernel<64, 1, 1, 2>;
      return tm32 ? conv_fwd_kernel<32, 1, 0, 2>
                  : conv_fwd_kernel<64, 1, 0, 2>;
    }
    if (stride == 1)
      return tm32 ? conv_fwd_kernel<32, 1, 1, 1>
                  : conv_fwd_kernel<64, 1, 1, 1>;
    if (stride == 2)
      return tm32 ? conv_fwd_kernel<32, 1, 2, 1>
                  : conv_fwd_kernel<64, 1, 2, 1>;
    return tm32 ? conv_fwd_kernel<32, 1, 0, 1> : conv_fwd_kernel<64, 1, 0, 1>;
  };
  const long long wgs64 = (long long)grid.x * grid.y * grid.z;
  if (wgs64 < 256 && M > CONV_TM) {
    dim3 grid32((M + 31) / 32, grid.y, grid.z);
    size_t lds32 = (size_t)4 * 32 * (64 + CONV_AP) * 2;
    hipLaunchKernelGGL(pick_fwd(true), grid32, dim3(256), lds32,
                       at::cuda::getCurrentCUDAStream(),
                       (const cvbf16*)xbuf.data_ptr(),
                       (const cvbf16*)wmat.data_ptr(), bptr,
                       (cvbf16*)out.data_ptr(), mbase.data_ptr<int>(),
                       koff.data_ptr<int>(), (int)M, (int)N, (int)K, (int)KP,
                       xbuf.stride(0), o_img, (int)act, (int)WO,
                       (int)stride, vH, vW, (int)vpt, (int)vpl, (int)vsv,
                       (int)oh0, (int)ow0, (int)ostep, WOf, o_chan);
    return out;
  }
  size_t lds = (size_t)4 * CONV_TM * (64 + CONV_AP) * 2;  // 4-buffer pipeline
  hipLaunchKernelGGL(pick_fwd(false), grid, dim3(256), lds,
                     at::cuda::getCurrentCUDAStream(),
                     (const cvbf16*)xbuf.data_ptr(),
                     (const cvbf16*)wmat.data_ptr(), bptr,
                     (cvbf16*)out.data_ptr(), mbase.data_ptr<int>(),
                     koff.data_ptr<int>(), (int)M, (int)N, (int)K, (int)KP,
                     xbuf.stride(0), o_img, (int)act, (int)WO,
                     (int)stride, vH, vW, (int)vpt, (int)vpl, (int)vsv,
                     (int)oh0, (int)ow0, (int)ostep, WOf, o_chan);
  return out;
}

torch::Tensor conv_wrw(torch::Tensor xbuf, torch::Tensor dy,
                       torch::Tensor mbase, torch::Tensor koff, int64_t N,
                       int64_t K, int64_t WO, bool mcontig, int64_t vm,
                       int64_t st, int64_t vpt, int64_t vpl, int64_t vsv) {
  CHECK_CUDA_CONTIG(xbuf);
  CHECK_CUDA_CONTIG(dy);
  const bool fp8 = xbuf.scalar_type() == torch::kByte;
  const int64_t B = xbuf.size(0);
  const int64_t M = dy.size(2) * dy.size(3);
  // pick pixel chunking so the grid fills the chip (~6 workgroups/CU).
  // Each (tap-tile, cout-tile, chunk) workgroup owns a disjoint 64x64 region
  // of its chunk's PARTIAL-SUM slice — plain stores, no atomics (atomicAdd
  // across ~40 chunks contends on the same dw cache lines and was measured
  // slower than the 9-chunk version it replaced). The slices are then
  // reduced with one sum(0). Total partial memory is bounded by the
  // workgroup target: <= 1536 * 64*64*4B = 25 MB.
  const int64_t tiles = ((K + 63) / 64) * ((N + 63) / 64) * B;
  int pix_chunks = (int)std::min<int64_t>(
      std::max<int64_t>(1536 / std::max<int64_t>(tiles, 1), 1),
      std::max<int64_t>(M / 256, 1));
  // every in-range (cout, tap) element of every slice is written exactly
  // once by its owning workgroup, so empty() needs no zero-fill.
  auto dwp = torch::empty({(int64_t)B * pix_chunks, N, K},
                          xbuf.options().dtype(torch::kFloat32));
  dim3 grid((K + 63) / 64, (N + 63) / 64, B * pix_chunks);
  if (fp8) {
    TORCH_CHECK(dy.scalar_type() == torch::kByte, "fp8 wrw: dy must be e4m3");
    size_t lds = (size_t)4 * 64 * (32 + CONV8_AP);
    hipLaunchKernelGGL(conv_wrw_fp8_kernel, grid, dim3(256), lds,
                       at::cuda::getCurrentCUDAStream(),
                       (const f8*)xbuf.data_ptr(), (const f8*)dy.data_ptr(),
                       dwp.data_ptr<float>(), mbase.data_ptr<int>(),
                       koff.data_ptr<int>(), (int)M, (int)N, (int)K,
                       xbuf.stride(0), dy.stride(0), pix_chunks, (int)WO,
                       (int)(mcontig ? 1 : 0));
  } else {
    size_t lds = (size_t)4 * 64 * (32 + CONV_AP) * 2;  // 2 tiles x dbuf
    const int vH = (int)xbuf.size(2), vW = (int)xbuf.size(3);
    using WrwKern = void (*)(const cvbf16*, const cvbf16*, float*,
                             const int*, const int*, int, int, int,
                             long long, long long, int, int, int, int, int,
                             int, int, int, int);
    WrwKern kern;
    if (!vm) {
      kern = conv_wrw_kernel<0, 0, 1>;
    } else {
      const bool gen = (WO < 8) || (st != 1 && st != 2);
      if (vsv == 2)
        kern = (!gen && st == 1) ? conv_wrw_kernel<1, 1, 2>
                                 : conv_wrw_kernel<1, 0, 2>;
      else if (gen)
        kern = conv_wrw_kernel<1, 0, 1>;
      else
        kern = (st == 1) ? conv_wrw_kernel<1, 1, 1> : conv_wrw_kernel<1, 2, 1>;
    }
    hipLaunchKernelGGL(kern, grid, dim3(256), lds,
                       at::cuda::getCurrentCUDAStream(),
                       (const cvbf16*)xbuf.data_ptr(),
                       (const cvbf16*)dy.data_ptr(), dwp.data_ptr<float>(),
                       mbase.data_ptr<int>(), koff.data_ptr<int>(), (int)M,
                       (int)N, (int)K, xbuf.stride(0), dy.stride(0),
                       pix_chunks, (int)WO, (int)(mcontig ? 1 : 0), (int)st,
                       vH, vW, (int)vpt, (int)vpl, (int)vsv);
  }
  if (B * pix_chunks == 1) return dwp.view({N, K});
  return dwp.sum(0);
}

torch::Tensor panel_gather(torch::Tensor src, torch::Tensor ktab) {
  CHECK_CUDA_CONTIG(src);
  CHECK_CUDA_CONTIG(ktab);
  const int64_t rows = src.size(0);
  const int64_t Kl = ktab.numel();
  const int64_t KPA = ((Kl + 63) & ~63) + CONV_AP;
  auto out = torch::empty({rows, KPA}, src.options().dtype(torch::kBFloat16));
  const long long total = rows * KPA;
  const int grid = (int)std::min<long long>((total + 255) / 256, 4096);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (src.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((panel_gather_kernel<float>), dim3(grid), dim3(256), 0,
                       stream, src.data_ptr<float>(), ktab.data_ptr<int>(),
                       (cvbf16*)out.data_ptr(), (int)rows, (int)Kl, (int)KPA,
                       (int)src.size(1));
  } else {
    TORCH_CHECK(src.scalar_type() == torch::kBFloat16, "panel src dtype");
    hipLaunchKernelGGL((panel_gather_kernel<cvbf16>), dim3(grid), dim3(256),
                       0, stream, (const cvbf16*)src.data_ptr(),
                       ktab.data_ptr<int>(), (cvbf16*)out.data_ptr(),
                       (int)rows, (int)Kl, (int)KPA, (int)src.size(1));
  }
  return out;
}

torch::Tensor act_bwd(torch::Tensor dy, torch::Tensor y, int64_t act) {
  CHECK_CUDA_CONTIG(dy);
  CHECK_CUDA_CONTIG(y);
  const long long n = dy.numel();
  auto out = torch::empty_like(y);  // y is bf16, same shape
  auto stream = at::cuda::getCurrentCUDAStream();
  const int grid = (int)std::min<long long>((n + 255) / 256, 4096);
  if (dy.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((act_bwd_kernel<float>), dim3(grid), dim3(256), 0,
                       stream, dy.data_ptr<float>(),
                       (const cvbf16*)y.data_ptr(), (cvbf16*)out.data_ptr(), n,
                       (int)act);
  } else {
    TORCH_CHECK(dy.scalar_type() == torch::kBFloat16, "act_bwd: fp32/bf16");
    hipLaunchKernelGGL((act_bwd_kernel<cvbf16>), dim3(grid), dim3(256), 0,
                       stream, (const cvbf16*)dy.data_ptr(),
                       (const cvbf16*)y.data_ptr(), (cvbf16*)out.data_ptr(), n,
                       (int)act);
  }
  return out;
}

torch::Tensor wmat_make(torch::Tensor w1, int64_t khw, int64_t mode) {
  CHECK_CUDA_CONTIG(w1);
  const int64_t N = w1.size(0), K = w1.size(1);
  const bool rot = (mode & 1) != 0;
  const int64_t rows = rot ? K / khw : N;
  const int64_t kout = rot ? N * khw : K;
  const int64_t KPA = ((kout + 63) & ~63) + CONV_AP;
  auto out = torch::empty({rows, KPA},
                          w1.options().dtype(torch::kBFloat16));
  const long long total = rows * KPA;
  const int grid = (int)std::min<long long>((total + 255) / 256, 4096);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (w1.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((wmat_make_kernel<float>), dim3(grid), dim3(256), 0,
                       stream, w1.data_ptr<float>(), (cvbf16*)out.data_ptr(),
                       (int)rows, (int)kout, (int)K, (int)khw, (int)KPA,
                       (int)mode);
  } else {
    TORCH_CHECK(w1.scalar_type() == torch::kBFloat16, "wmat_make: fp32/bf16");
    hipLaunchKernelGGL((wmat_make_kernel<cvbf16>), dim3(grid), dim3(256), 0,
                       stream, (const cvbf16*)w1.data_ptr(),
                       (cvbf16*)out.data_ptr(), (int)rows, (int)kout, (int)K,
                       (int)khw, (int)KPA, (int)mode);
  }
  return out;
}

}  // namespace dsin
