// Fused elementwise/reduction kernels for the eager-torch remainder
// (round-1 verdict item 6): heatmap construction +
// bottleneck masking, L1-distortion partial sums, and the H_real/H_mask
// rate-term sums. Each replaces a 3-8 kernel torch chain (with full-tensor
// temporaries) by one kernel + one ordered partial-sum reduce, and every
// reduction is deterministic (plain per-block partial stores, host-side
// at::sum in fixed order).

#include "common.h"

namespace dsin {

using fubf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(2))) float fu_f32x2;

// ---------------------------------------------------------------- heatmap
// Reference src/autoencoder_imgcomp.py:172-201:
//   heatmap2D = sigmoid(b[:,0]) * C;  h3[.,c] = clamp(heatmap2D - c, 0, 1)
//   z = h3 * b[:, 1:]
// One thread per (n, h, w): computes the sigmoid once, loops the C channels.
template <typename T>
__global__ void heatmap_mask_fwd_kernel(const T* __restrict__ b,
                                        T* __restrict__ z,
                                        T* __restrict__ h3,
                                        int C, long long HW, int N) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = (long long)N * HW;
  const long long gstride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += gstride) {
    const long long n = i / HW, hw = i % HW;
    const T* bn = b + (n * (C + 1)) * HW + hw;
    const float h2 = (1.f / (1.f + __expf(-(float)bn[0]))) * (float)C;
    T* zn = z + (n * C) * HW + hw;
    T* hn = h3 + (n * C) * HW + hw;
    for (int c = 0; c < C; ++c) {
      const float h = fminf(fmaxf(h2 - (float)c, 0.f), 1.f);
      hn[(long long)c * HW] = (T)h;
      zn[(long long)c * HW] = (T)(h * (float)bn[(long long)(c + 1) * HW]);
    }
  }
}

// backward: g_b[:,0] = C * s * (1 - s) * sum_c (g_z[c]*b[c+1] + g_h3[c])
//                     * 1{0 <= h2 - c <= 1}   (torch clamp grad semantics)
//           g_b[:,c+1] = g_z[c] * h3[c]
template <typename T>
__global__ void heatmap_mask_bwd_kernel(const T* __restrict__ b,
                                        const T* __restrict__ gz,
                                        const T* __restrict__ gh3,
                                        T* __restrict__ gb,
                                        int C, long long HW, int N) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = (long long)N * HW;
  const long long gstride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += gstride) {
    const long long n = i / HW, hw = i % HW;
    const T* bn = b + (n * (C + 1)) * HW + hw;
    const T* gzn = gz + (n * C) * HW + hw;
    const T* ghn = gh3 ? gh3 + (n * C) * HW + hw : nullptr;
    T* gbn = gb + (n * (C + 1)) * HW + hw;
    const float s = 1.f / (1.f + __expf(-(float)bn[0]));
    const float h2 = s * (float)C;
    float acc = 0.f;
    for (int c = 0; c < C; ++c) {
      const float v = h2 - (float)c;
      const float h = fminf(fmaxf(v, 0.f), 1.f);
      const float gzc = (float)gzn[(long long)c * HW];
      gbn[(long long)(c + 1) * HW] = (T)(gzc * h);
      if (v >= 0.f && v <= 1.f) {
        float g = gzc * (float)bn[(long long)(c + 1) * HW];
        if (ghn) g += (float)ghn[(long long)c * HW];
        acc += g;
      }
    }
    gbn[0] = (T)(acc * (float)C * s * (1.f - s));
  }
}

// ---------------------------------------------------------------- L1 mean
// Per-image |y - x| partial sums: parts[n][s] (plain stores, one ordered
// at::sum per image on the host). x and y may differ in dtype.
template <typename TX, typename TY>
__global__ void l1_part_kernel(const TX* __restrict__ x,
                               const TY* __restrict__ y,
                               float* __restrict__ parts,
                               long long CHW, int S) {
  const int n = blockIdx.y;
  const int s = blockIdx.x;
  const TX* xn = x + (long long)n * CHW;
  const TY* yn = y + (long long)n * CHW;
  float a = 0.f;
  for (long long i = (long long)s * blockDim.x + threadIdx.x; i < CHW;
       i += (long long)S * blockDim.x)
    a += fabsf((float)yn[i] - (float)xn[i]);
  a = wave_reduce_sum(a);
  __shared__ float red[4];
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = a;
  __syncthreads();
  if (threadIdx.x == 0)
    parts[(long long)n * S + s] = red[0] + red[1] + red[2] + red[3];
}

// backward: g per image scalar; gx = -sign(y-x)*g, gy = +sign(y-x)*g
template <typename TX, typename TY>
__global__ void l1_bwd_kernel(const TX* __restrict__ x,
                              const TY* __restrict__ y,
                              const float* __restrict__ g,  // (N,)
                              TX* __restrict__ gx,          // nullable
                              TY* __restrict__ gy,          // nullable
                              long long CHW, int N) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = (long long)N * CHW;
  const long long gstride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += gstride) {
    const long long n = i / CHW;
    const float d = (float)y[i] - (float)x[i];
    const float sg = d > 0.f ? 1.f : (d < 0.f ? -1.f : 0.f);
    const float gv = sg * g[n];
    if (gy) gy[i] = (TY)gv;
    if (gx) gx[i] = (TX)(-gv);
  }
}

// ------------------------------------------------------------- rate terms
// parts[s] = (sum bc, sum bc*heat) over slice s — one pass instead of two
// full-tensor reads (H_real and H_mask, Distortions_imgcomp.py:118-124).
__global__ void hterms_part_kernel(const float* __restrict__ bc,
                                   const fubf16* __restrict__ heat,
                                   float* __restrict__ parts,  // (S, 2)
                                   long long n, int S) {
  const int s = blockIdx.x;
  float a = 0.f, b = 0.f;
  for (long long i = (long long)s * blockDim.x + threadIdx.x; i < n;
       i += (long long)S * blockDim.x) {
    const float v = bc[i];
    a += v;
    b += v * (float)heat[i];
  }
  a = wave_reduce_sum(a);
  b = wave_reduce_sum(b);
  __shared__ float red[2][4];
  if ((threadIdx.x & 63) == 0) {
    red[0][threadIdx.x >> 6] = a;
    red[1][threadIdx.x >> 6] = b;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    parts[(long long)s * 2] = red[0][0] + red[0][1] + red[0][2] + red[0][3];
    parts[(long long)s * 2 + 1] =
        red[1][0] + red[1][1] + red[1][2] + red[1][3];
  }
}

// backward: g_bc = g1/n + g2/n * heat ; g_heat = g2/n * bc
__global__ void hterms_bwd_kernel(const float* __restrict__ bc,
                                  const fubf16* __restrict__ heat,
                                  const float* __restrict__ g2v,  // (2,)
                                  float* __restrict__ gbc,
                                  fubf16* __restrict__ gheat,
                                  long long n, float invn) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long gstride = (long long)gridDim.x * blockDim.x;
  const float g1 = g2v[0] * invn, g2 = g2v[1] * invn;
  for (; i < n; i += gstride) {
    gbc[i] = g1 + g2 * (float)heat[i];
    if (gheat) gheat[i] = __float2bfloat16(g2 * bc[i]);
  }
}

// ------------------------------------------------------------------ hosts

std::vector<torch::Tensor> heatmap_mask_fwd(torch::Tensor b) {
  CHECK_CUDA_CONTIG(b);
  const int N = (int)b.size(0), C1 = (int)b.size(1);
  const long long HW = (long long)b.size(2) * b.size(3);
  auto z = torch::empty({N, C1 - 1, b.size(2), b.size(3)}, b.options());
  auto h3 = torch::empty_like(z);
  const long long total = (long long)N * HW;
  const int grid = (int)std::min<long long>((total + 255) / 256, 4096);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (b.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL((heatmap_mask_fwd_kernel<float>), dim3(grid),
                       dim3(256), 0, stream, b.data_ptr<float>(),
                       z.data_ptr<float>(), h3.data_ptr<float>(), C1 - 1, HW,
                       N);
  } else {
    TORCH_CHECK(b.scalar_type() == torch::kBFloat16, "heatmap: fp32/bf16");
    hipLaunchKernelGGL((heatmap_mask_fwd_kernel<fubf16>), dim3(grid),
                       dim3(256), 0, stream, (const fubf16*)b.data_ptr(),
                       (fubf16*)z.data_ptr(), (fubf16*)h3.data_ptr(), C1 - 1,
                       HW, N);
  }
  return {z, h3};
}

torch::Tensor heatmap_mask_bwd(torch::Tensor b, torch::Tensor gz,
                               c10::optional<torch::Tensor> gh3) {
  CHECK_CUDA_CONTIG(b);
  CHECK_CUDA_CONTIG(gz);
  const int N = (int)b.size(0), C1 = (int)b.size(1);
  const long long HW = (long long)b.size(2) * b.size(3);
  auto gb = torch::empty_like(b);
  const long long total = (long long)N * HW;
  const int grid = (int)std::min<long long>((total + 255) / 256, 4096);
  auto stream = at::cuda::getCurrentCUDAStream();
  if (b.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(
        (heatmap_mask_bwd_kernel<float>), dim3(grid), dim3(256), 0, stream,
        b.data_ptr<float>(), gz.data_ptr<float>(),
        gh3.has_value() ? gh3->data_ptr<float>() : nullptr,
        gb.data_ptr<float>(), C1 - 1, HW, N);
  } else {
    hipLaunchKernelGGL(
        (heatmap_mask_bwd_kernel<fubf16>), dim3(grid), dim3(256), 0, stream,
        (const fubf16*)b.data_ptr(), (const fubf16*)gz.data_ptr(),
        gh3.has_value() ? (const fubf16*)gh3->data_ptr() : nullptr,
        (fubf16*)gb.data_ptr(), C1 - 1, HW, N);
  }
  return gb;
}

static int _l1_slices(long long CHW) {
  return (int)std::min<long long>(std::max<long long>(CHW / (256 * 16), 1),
                                  64);
}

torch::Tensor l1_part(torch::Tensor x, torch::Tensor y) {
  CHECK_CUDA_CONTIG(x);
  CHECK_CUDA_CONTIG(y);
  const int N = (int)x.size(0);
  const long long CHW = x.numel() / N;
  const int S = _l1_slices(CHW);
  auto parts = torch::empty({N, S}, x.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentCUDAStream();
  const bool xf = x.scalar_type() == torch::kFloat32;
  const bool yf = y.scalar_type() == torch::kFloat32;
#define L1P(TX, TY)                                                         \
  hipLaunchKernelGGL((l1_part_kernel<TX, TY>), dim3(S, N), dim3(256), 0,    \
                     stream, (const TX*)x.data_ptr(),                       \
                     (const TY*)y.data_ptr(), parts.data_ptr<float>(), CHW, \
                     S)
  if (xf && yf) L1P(float, float);
  else if (xf && !yf) L1P(float, fubf16);
  else if (!xf && yf) L1P(fubf16, float);
  else L1P(fubf16, fubf16);
#undef L1P
  return parts;
}

std::vector<torch::Tensor> l1_bwd(torch::Tensor x, torch::Tensor y,
                                  torch::Tensor g, bool need_gx,
                                  bool need_gy) {
  const int N = (int)x.size(0);
  const long long CHW = x.numel() / N;
  auto gx = need_gx ? torch::empty_like(x) : torch::Tensor();
  auto gy = need_gy ? torch::empty_like(y) : torch::Tensor();
  const long long total = x.numel();
  const int grid = (int)std::min<long long>((total + 255) / 256, 4096);
  auto stream = at::cuda::getCurrentCUDAStream();
  const bool xf = x.scalar_type() == torch::kFloat32;
  const bool yf = y.scalar_type() == torch::kFloat32;
#define L1B(TX, TY)                                                          \
  hipLaunchKernelGGL((l1_bwd_kernel<TX, TY>), dim3(grid), dim3(256), 0,      \
                     stream, (const TX*)x.data_ptr(),                        \
                     (const TY*)y.data_ptr(), g.data_ptr<float>(),           \
                     need_gx ? (TX*)gx.data_ptr() : nullptr,                 \
                     need_gy ? (TY*)gy.data_ptr() : nullptr, CHW, N)
  if (xf && yf) L1B(float, float);
  else if (xf && !yf) L1B(float, fubf16);
  else if (!xf && yf) L1B(fubf16, float);
  else L1B(fubf16, fubf16);
#undef L1B
  return {gx, gy};
}

torch::Tensor hterms_part(torch::Tensor bc, torch::Tensor heat) {
  CHECK_CUDA_CONTIG(bc);
  CHECK_CUDA_CONTIG(heat);
  TORCH_CHECK(bc.scalar_type() == torch::kFloat32 &&
                  heat.scalar_type() == torch::kBFloat16,
              "hterms: bc fp32, heat bf16");
  const long long n = bc.numel();
  const int S = _l1_slices(n);
  auto parts = torch::empty({S, 2}, bc.options());
  hipLaunchKernelGGL(hterms_part_kernel, dim3(S), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), bc.data_ptr<float>(),
                     (const fubf16*)heat.data_ptr(), parts.data_ptr<float>(),
                     n, S);
  return parts;
}

std::vector<torch::Tensor> hterms_bwd(torch::Tensor bc, torch::Tensor heat,
                                      torch::Tensor g2, bool need_gheat) {
  const long long n = bc.numel();
  auto gbc = torch::empty_like(bc);
  auto gheat = need_gheat ? torch::empty_like(heat) : torch::Tensor();
  const int grid = (int)std::min<long long>((n + 255) / 256, 4096);
  hipLaunchKernelGGL(hterms_bwd_kernel, dim3(grid), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), bc.data_ptr<float>(),
                     (const fubf16*)heat.data_ptr(), g2.data_ptr<float>(),
                     gbc.data_ptr<float>(),
                     need_gheat ? (fubf16*)gheat.data_ptr() : nullptr, n,
                     1.f / (float)n);
  return {gbc, gheat};
}

}  // namespace dsin
