// Streaming side-information NCC search for CDNA4 (gfx950) — host wrapper.
//
// Reference semantics (/root/reference/src/siFinder.py:7-135,
// src/siFull_img.py:5-68, mask from src/AE.py:49,193-220): every
// non-overlapping (ph, pw) patch of the decoded image x is Pearson-correlated
// against every location of the decoded side image y (both fixed-normalized
// and H1H2H3-decorrelated), the correlation is weighted by a per-patch
// Gaussian location prior, the argmax location is found, and the winning
// patches are gathered FROM THE ORIGINAL y and scattered into y_syn.
//
// The reference materializes the (Hc, Wc, P) correlation volume and an
// equally-sized mask constant (~722 MB each at 320x960). This implementation
// streams: the correlation xy[p,i,j] is an implicit GEMM
// (M = P patches, N = Hc*Wc locations, K = 3*ph*pw) on bf16 MFMA
// (16x16x32, fp32 accumulate); Pearson normalization, the Gaussian prior
// (evaluated inline, never materialized) and a packed-u64 atomic argmax with
// TF tie semantics (smallest index wins) live in the epilogue. Device code in
// ncc_kernels.h (torch-free, compilable standalone for .s inspection).

#include "common.h"
#include "ncc_kernels.h"

namespace dsin {

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> ncc_search(
    torch::Tensor x_dec, torch::Tensor y_dec, torch::Tensor y_orig,
    int64_t ph_, int64_t pw_, bool use_mask) {
  CHECK_CUDA_CONTIG(x_dec);
  CHECK_CUDA_CONTIG(y_dec);
  CHECK_CUDA_CONTIG(y_orig);
  TORCH_CHECK(x_dec.dim() == 3 && x_dec.size(0) == 3, "x_dec must be (3,H,W)");
  TORCH_CHECK(x_dec.sizes() == y_dec.sizes() && x_dec.sizes() == y_orig.sizes(),
              "x/y size mismatch (equal sizes required)");
  const int ph = (int)ph_, pw = (int)pw_;
  const int H = (int)x_dec.size(1), W = (int)x_dec.size(2);
  TORCH_CHECK(H % ph == 0 && W % pw == 0, "image must tile by the patch size");
  const int gh = H / ph, gw = W / pw, P = gh * gw;
  const int Hc = H - ph + 1, Wc = W - pw + 1;
  const int K = 3 * ph * pw;
  const int KP = (K + 15) & ~15;
  const int YR = NCC_TI + ph - 1, YC = NCC_TJ + pw - 1, YCP = (YC + 8) & ~7;
  // v2 (pw % 8 == 0, the default geometry): k-sliced LDS A-tile + direct
  // global B loads — see ncc_main_v2_kernel
  const bool v2 = (pw % 8) == 0;
  const size_t lds =
      v2 ? (size_t)2 * NCC_TP * (NCC_SL + NCC_APAD) * 2 + (size_t)5 * NCC_TP * 4
         : (size_t)NCC_TP * (KP + NCC_APAD) * 2 + (size_t)3 * YR * YCP * 2 +
               (size_t)((KP + 7) & ~7) * 2 + (size_t)5 * NCC_TP * 4;
  TORCH_CHECK(lds <= 160 * 1024, "patch size too large for LDS tiling: ", lds);
  TORCH_CHECK(v2 || 3 * YR * YCP < 65536, "y-window exceeds u16 offset range");

  auto optsF = x_dec.options();
  auto stream = at::cuda::getCurrentCUDAStream();

  auto aoffs = torch::empty({K}, optsF.dtype(torch::kUInt32));
  auto koffs = torch::empty({K}, optsF.dtype(torch::kUInt16));
  hipLaunchKernelGGL(ncc_offsets_kernel, grid1d(K, 256), dim3(256), 0, stream,
                     (unsigned int*)aoffs.data_ptr(),
                     (unsigned short*)koffs.data_ptr(), H, W, ph, pw, YR, YCP,
                     K);

  auto t_x = torch::empty({3, H, W}, optsF.dtype(torch::kBFloat16));
  // +192 elements of tail slack: the v2 kernel's invalid-column vector
  // loads may overshoot the last row (values discarded)
  auto t_y = torch::empty({(int64_t)3 * H * W + 192},
                          optsF.dtype(torch::kBFloat16));
  int64_t hw = (int64_t)H * W;
  hipLaunchKernelGGL(transform_kernel, grid1d(hw, 256), dim3(256), 0, stream,
                     x_dec.data_ptr<float>(), (ncbf16*)t_x.data_ptr(), hw);
  hipLaunchKernelGGL(transform_kernel, grid1d(hw, 256), dim3(256), 0, stream,
                     y_dec.data_ptr<float>(), (ncbf16*)t_y.data_ptr(), hw);

  auto psum = torch::empty({P}, optsF);
  auto psum2 = torch::empty({P}, optsF);
  hipLaunchKernelGGL(patch_stats_kernel, dim3(P), dim3(64), 0, stream,
                     (const ncbf16*)t_x.data_ptr(),
                     (const unsigned int*)aoffs.data_ptr(),
                     psum.data_ptr<float>(), psum2.data_ptr<float>(), H, W, ph,
                     pw, gw, K);

  auto s1 = torch::empty({(int64_t)H * Wc}, optsF);
  auto s2 = torch::empty({(int64_t)H * Wc}, optsF);
  auto sy = torch::empty({(int64_t)Hc * Wc}, optsF);
  auto sy2 = torch::empty({(int64_t)Hc * Wc}, optsF);
  hipLaunchKernelGGL(ysum_row_kernel, grid1d((int64_t)H * Wc, 256), dim3(256),
                     0, stream, (const ncbf16*)t_y.data_ptr(),
                     s1.data_ptr<float>(), s2.data_ptr<float>(), H, W, pw, Wc);
  hipLaunchKernelGGL(ysum_col_kernel, grid1d((int64_t)Hc * Wc, 256), dim3(256),
                     0, stream, s1.data_ptr<float>(), s2.data_ptr<float>(),
                     sy.data_ptr<float>(), sy2.data_ptr<float>(), Hc, Wc, ph);

  auto best = torch::zeros({P}, optsF.dtype(torch::kInt64));  // u64 keys
  const int tj = v2 ? NCC_TJ2 : NCC_TJ;
  dim3 grid((Wc + tj - 1) / tj, (Hc + NCC_TI - 1) / NCC_TI,
            (P + NCC_TP - 1) / NCC_TP);
  if (v2) {
    hipLaunchKernelGGL(ncc_main_v2_kernel, grid, dim3(256), lds, stream,
                       (const ncbf16*)t_x.data_ptr(),
                       (const ncbf16*)t_y.data_ptr(),
                       (const unsigned int*)aoffs.data_ptr(),
                       psum.data_ptr<float>(), psum2.data_ptr<float>(),
                       sy.data_ptr<float>(), sy2.data_ptr<float>(),
                       (unsigned long long*)best.data_ptr(), H, W, ph, pw,
                       gw, P, Hc, Wc, use_mask ? 1 : 0);
  } else {
    hipLaunchKernelGGL(ncc_main_kernel, grid, dim3(256), lds, stream,
                       (const ncbf16*)t_x.data_ptr(),
                       (const ncbf16*)t_y.data_ptr(),
                       (const unsigned int*)aoffs.data_ptr(),
                       (const unsigned short*)koffs.data_ptr(),
                       psum.data_ptr<float>(), psum2.data_ptr<float>(),
                       sy.data_ptr<float>(), sy2.data_ptr<float>(),
                       (unsigned long long*)best.data_ptr(), H, W, ph, pw,
                       gw, P, Hc, Wc, use_mask ? 1 : 0);
  }

  auto y_syn = torch::empty_like(y_orig);
  auto rows = torch::empty({P}, optsF.dtype(torch::kInt64));
  auto cols = torch::empty({P}, optsF.dtype(torch::kInt64));
  hipLaunchKernelGGL(scatter_kernel, dim3(P), dim3(256), 0, stream,
                     (const unsigned long long*)best.data_ptr(),
                     y_orig.data_ptr<float>(), y_syn.data_ptr<float>(),
                     (long long*)rows.data_ptr<int64_t>(),
                     (long long*)cols.data_ptr<int64_t>(), H, W, ph, pw, gw, P,
                     Wc);
  return {y_syn, rows, cols};
}

// MFMA layout self-check used by the GPU tests: C = A(16x32) @ B(32x16), bf16
// inputs, one wave.
__global__ void mfma_selftest_kernel(const float* __restrict__ A,
                                     const float* __restrict__ B,
                                     float* __restrict__ C) {
  int lane = threadIdx.x & 63;
  bf16x8 a, b;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    int k = (lane >> 4) * 8 + e;
    ncbf16 av = nf2b(A[(lane & 15) * 32 + k]);   // A[row][k]
    ncbf16 bv = nf2b(B[k * 16 + (lane & 15)]);   // B[k][col]
    a[e] = *reinterpret_cast<__bf16*>(&av);
    b[e] = *reinterpret_cast<__bf16*>(&bv);
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    int row = (lane >> 4) * 4 + reg, col = lane & 15;
    C[row * 16 + col] = acc[reg];
  }
}

torch::Tensor mfma_selftest(torch::Tensor A, torch::Tensor B) {
  CHECK_CUDA_CONTIG(A);
  CHECK_CUDA_CONTIG(B);
  auto C = torch::empty({16, 16}, A.options());
  hipLaunchKernelGGL(mfma_selftest_kernel, dim3(1), dim3(64), 0,
                     at::cuda::getCurrentCUDAStream(), A.data_ptr<float>(),
                     B.data_ptr<float>(), C.data_ptr<float>());
  return C;
}

}  // namespace dsin

namespace dsin {
// 32x32x16 MFMA layout self-check: C = A(32x16) @ B(16x32), one wave.
__global__ void mfma32_selftest_kernel(const float* __restrict__ A,
                                       const float* __restrict__ B,
                                       float* __restrict__ C) {
  typedef __attribute__((ext_vector_type(8))) __bf16 b8;
  typedef __attribute__((ext_vector_type(16))) float f16v;
  int lane = threadIdx.x & 63;
  b8 a, b;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    int k = (lane >> 5) * 8 + e;
    ncbf16 av = nf2b(A[(lane & 31) * 16 + k]);   // A[row][k]
    ncbf16 bv = nf2b(B[k * 32 + (lane & 31)]);   // B[k][col]
    a[e] = *reinterpret_cast<__bf16*>(&av);
    b[e] = *reinterpret_cast<__bf16*>(&bv);
  }
  f16v acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
    int col = lane & 31;
    C[row * 32 + col] = acc[reg];
  }
}

torch::Tensor mfma32_selftest(torch::Tensor A, torch::Tensor B) {
  CHECK_CUDA_CONTIG(A);
  CHECK_CUDA_CONTIG(B);
  auto C = torch::empty({32, 32}, A.options());
  hipLaunchKernelGGL(mfma32_selftest_kernel, dim3(1), dim3(64), 0,
                     at::cuda::getCurrentCUDAStream(), A.data_ptr<float>(),
                     B.data_ptr<float>(), C.data_ptr<float>());
  return C;
}
}  // namespace dsin
