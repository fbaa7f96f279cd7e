#include "hip/hip_runtime.h"
// Fused pad / zero-stuff / cast kernel: builds the conv input buffer
// (bf16, zero border, optional stride-S zero-stuffing for transposed convs)
// in ONE pass from an fp32 or bf16 NCHW tensor — replacing the
// zeros-fill + cast + interior-copy kernel chain per conv call.
// The output buffer carries 16 elements of tail slack for the conv staging
// vector reads (see ops/conv.py).

#include "common_hip.h"
#include "conv_fp8.h"

namespace dsin {

// one block row-group per (b, c, output row); threads stride the row.
// Row-oriented addressing keeps the per-element index math 32-bit and
// branch-uniform (the old flat-index version spent most of its time in
// 64-bit div/mod per element: 13 us for a 4 MB buffer).
template <typename T>
__global__ void pad_stuff_kernel(const T* __restrict__ x,
                                 bf16* __restrict__ out,
                                 int C, int H, int W, int Hp, int Wp,
                                 int pt, int pl, int stride,
                                 long long n_img_out, long long n_img_in,
                                 int B) {
  const int row = blockIdx.x;          // (b*C + c)*Hp + i
  const int i = row % Hp;
  const int bc = row / Hp;
  bf16* orow = out + (long long)bc * Hp * Wp + (long long)i * Wp;
  const int ii = i - pt;
  const bool rowin = ii >= 0 && (stride == 1 || ii % stride == 0) &&
                     ii / stride < H;
  if (!rowin) {
    for (int j = threadIdx.x; j < Wp; j += blockDim.x) orow[j] = f2b(0.f);
    return;
  }
  const T* xrow = x + ((long long)bc * H + ii / stride) * W;
  for (int j = threadIdx.x; j < Wp; j += blockDim.x) {
    const int jj = j - pl;
    float v = 0.f;
    if (jj >= 0 && (stride == 1 || jj % stride == 0) && jj / stride < W)
      v = (float)xrow[jj / stride];
    orow[j] = f2b(v);
  }
}

torch::Tensor pad_stuff(torch::Tensor x, int64_t pt, int64_t pb, int64_t pl,
                        int64_t pr, int64_t stride, bool fp8) {
  CHECK_CUDA_CONTIG(x);
  TORCH_CHECK(x.dim() == 4, "pad_stuff expects NCHW");
  const int B = (int)x.size(0), C = (int)x.size(1);
  const int H = (int)x.size(2), W = (int)x.size(3);
  const int Hs = (H - 1) * (int)stride + 1, Ws = (W - 1) * (int)stride + 1;
  const int Hp = Hs + (int)(pt + pb), Wp = Ws + (int)(pl + pr);
  const long long n_img = (long long)C * Hp * Wp;
  auto store = torch::empty({(int64_t)B * n_img + 16},
                            x.options().dtype(fp8 ? torch::kByte
                                                  : torch::kBFloat16));
  auto out = store.narrow(0, 0, B * n_img).view({B, C, Hp, Wp});
  long long total = (long long)B * n_img;
  int grid = (int)std::min<long long>((total + 255) / 256, 8192);
  const int rows = B * C * Hp;
  const int rthreads = Wp >= 256 ? 256 : (Wp >= 128 ? 128 : 64);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  if (fp8) {
    if (x.scalar_type() == torch::kFloat32) {
      hipLaunchKernelGGL((pad_stuff_fp8_kernel<float>), dim3(grid), dim3(256),
                         0, stream, x.data_ptr<float>(), (f8*)out.data_ptr(),
                         C, H, W, Hp, Wp, (int)pt, (int)pl, (int)stride, n_img,
                         (long long)C * H * W, B);
    } else if (x.scalar_type() == torch::kBFloat16) {
      hipLaunchKernelGGL((pad_stuff_fp8_kernel<bf16>), dim3(grid), dim3(256),
                         0, stream, (const bf16*)x.data_ptr(),
                         (f8*)out.data_ptr(), C, H, W, Hp, Wp, (int)pt,
                         (int)pl, (int)stride, n_img, (long long)C * H * W, B);
    } else {
      TORCH_CHECK(false, "pad_stuff fp8: fp32 or bf16 input only");
    }
    return out;
  }
  if (x.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((pad_stuff_kernel<float>), dim3(rows), dim3(rthreads),
                       0, stream, x.data_ptr<float>(), (bf16*)out.data_ptr(),
                       C, H, W, Hp, Wp, (int)pt, (int)pl, (int)stride, n_img,
                       (long long)C * H * W, B);
  } else if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((pad_stuff_kernel<bf16>), dim3(rows), dim3(rthreads),
                       0, stream, (const bf16*)x.data_ptr(),
                       (bf16*)out.data_ptr(), C, H, W, Hp, Wp, (int)pt,
                       (int)pl, (int)stride, n_img, (long long)C * H * W, B);
  } else {
    TORCH_CHECK(false, "pad_stuff: fp32 or bf16 only");
  }
  return out;
}

}  // namespace dsin
