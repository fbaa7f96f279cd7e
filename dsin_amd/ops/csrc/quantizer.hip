// Fused soft-to-hard scalar quantizer, forward + analytic backward.
//
// Semantics mirror /root/reference/src/quantizer_imgcomp.py:37-100 plus the
// straight-through combine at src/autoencoder_imgcomp.py:131-134:
//   phi    = softmax(-sigma * (x - c)^2) over the L centers
//   qsoft  = sum_l phi_l c_l ;  symbols = argmin_l (x - c_l)^2 ; qhard = c[sym]
//   qbar   = qsoft + sg(qhard - qsoft)   (forward value == qhard)
// Backward (the qsoft path only): with u_l = -sigma (x-c_l)^2,
//   d qsoft/dx   = sum_l c_l phi_l u'_l - qsoft * sum_l phi_l u'_l,
//                  u'_l = -2 sigma (x - c_l)
//   d qsoft/dc_j = phi_j + 2 sigma (x - c_j) phi_j (c_j - qsoft)
// Centers gradients: per-wave LDS partials summed in fixed order, plain
// per-block stores, one ordered host-side sum — bitwise-deterministic.
//
// One thread per element; L <= 16 centers live in registers (read through
// a small constant-ish global array; L=6 in the shipped config).

#include "common.h"

namespace dsin {

constexpr int MAX_L = 16;

__global__ void quantize_fwd_kernel(const float* __restrict__ x,
                                    const float* __restrict__ centers,
                                    float* __restrict__ qbar,
                                    int64_t* __restrict__ symbols,
                                    int L, float sigma, int64_t n) {
  float c[MAX_L];
#pragma unroll 8
  for (int l = 0; l < L; ++l) c[l] = centers[l];
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float xi = x[i];
    float dmin = 1e30f;
    int sym = 0;
#pragma unroll 8
    for (int l = 0; l < L; ++l) {
      float d = (xi - c[l]) * (xi - c[l]);
      if (d < dmin) { dmin = d; sym = l; }
    }
    // softmax over -sigma*d, stabilized by dmin (max of -sigma*d)
    float denom = 0.f, num = 0.f;
#pragma unroll 8
    for (int l = 0; l < L; ++l) {
      float d = (xi - c[l]) * (xi - c[l]);
      float p = __expf(-sigma * (d - dmin));
      denom += p;
      num += p * c[l];
    }
    float qsoft = num / denom;
    float qhard = c[sym];
    qbar[i] = qsoft + (qhard - qsoft);  // == qhard; written as the ref computes
    symbols[i] = sym;
  }
}

__global__ void quantize_bwd_kernel(const float* __restrict__ g,
                                    const float* __restrict__ x,
                                    const float* __restrict__ centers,
                                    float* __restrict__ gx,
                                    float* __restrict__ gc,  // (L,) accumulated
                                    int L, float sigma, int64_t n) {
  float c[MAX_L];
  float gcl[MAX_L];
#pragma unroll 8
  for (int l = 0; l < L; ++l) { c[l] = centers[l]; gcl[l] = 0.f; }
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float xi = x[i];
    float gi = g[i];
    float dmin = 1e30f;
#pragma unroll 8
    for (int l = 0; l < L; ++l) {
      float d = (xi - c[l]) * (xi - c[l]);
      dmin = fminf(dmin, d);
    }
    float phi[MAX_L];
    float denom = 0.f, qsoft = 0.f;
#pragma unroll 8
    for (int l = 0; l < L; ++l) {
      float d = (xi - c[l]) * (xi - c[l]);
      phi[l] = __expf(-sigma * (d - dmin));
      denom += phi[l];
    }
    float inv = 1.f / denom;
#pragma unroll 8
    for (int l = 0; l < L; ++l) { phi[l] *= inv; qsoft += phi[l] * c[l]; }

    float s_cw = 0.f, s_w = 0.f;
#pragma unroll 8
    for (int l = 0; l < L; ++l) {
      float w = -2.f * sigma * (xi - c[l]);  // du_l/dx
      s_cw += c[l] * phi[l] * w;
      s_w += phi[l] * w;
    }
    gx[i] = gi * (s_cw - qsoft * s_w);
#pragma unroll 8
    for (int l = 0; l < L; ++l) {
      float dqc = phi[l] + 2.f * sigma * (xi - c[l]) * phi[l] * (c[l] - qsoft);
      gcl[l] += gi * dqc;
    }
  }
  // DETERMINISTIC centers-grad reduction: per-wave partials in LDS, summed
  // in fixed wave order by one thread, then a plain per-block store; the
  // host reduces the (grid, L) partials with one ordered sum. (fp32
  // atomicAdd — across blocks or across the 4 waves of a block — has a
  // run-varying order and was one of the two nondeterminism sources.)
  __shared__ float red[4][MAX_L];
  const int wv = threadIdx.x >> 6;
#pragma unroll 8
  for (int l = 0; l < L; ++l) {
    float v = wave_reduce_sum(gcl[l]);
    if ((threadIdx.x & 63) == 0) red[wv][l] = v;
  }
  __syncthreads();
  if (threadIdx.x < L)
    gc[(int64_t)blockIdx.x * L + threadIdx.x] =
        red[0][threadIdx.x] + red[1][threadIdx.x] + red[2][threadIdx.x] +
        red[3][threadIdx.x];
}

std::tuple<torch::Tensor, torch::Tensor> quantize_fwd(torch::Tensor x,
                                                      torch::Tensor centers,
                                                      double sigma) {
  CHECK_CUDA_CONTIG(x);
  CHECK_CUDA_CONTIG(centers);
  TORCH_CHECK(x.scalar_type() == torch::kFloat32, "quantize: x must be fp32");
  int L = centers.numel();
  TORCH_CHECK(L <= MAX_L, "quantize: at most ", MAX_L, " centers");
  auto qbar = torch::empty_like(x);
  auto symbols = torch::empty(x.sizes(), x.options().dtype(torch::kInt64));
  int64_t n = x.numel();
  int block = 256;
  int grid = std::min<int64_t>((n + block - 1) / block, 4096);
  hipLaunchKernelGGL(quantize_fwd_kernel, dim3(grid), dim3(block), 0,
                     at::cuda::getCurrentCUDAStream(),
                     x.data_ptr<float>(), centers.data_ptr<float>(),
                     qbar.data_ptr<float>(), symbols.data_ptr<int64_t>(),
                     L, (float)sigma, n);
  return {qbar, symbols};
}

std::tuple<torch::Tensor, torch::Tensor> quantize_bwd(torch::Tensor g,
                                                      torch::Tensor x,
                                                      torch::Tensor centers,
                                                      double sigma) {
  CHECK_CUDA_CONTIG(g);
  CHECK_CUDA_CONTIG(x);
  CHECK_CUDA_CONTIG(centers);
  int L = centers.numel();
  auto gx = torch::empty_like(x);
  int64_t n = x.numel();
  int block = 256;
  int grid = std::min<int64_t>((n + block - 1) / block, 2048);
  // (grid, L) per-block partials, reduced with one ordered sum
  auto gcp = torch::empty({grid, L}, centers.options());
  hipLaunchKernelGGL(quantize_bwd_kernel, dim3(grid), dim3(block), 0,
                     at::cuda::getCurrentCUDAStream(),
                     g.data_ptr<float>(), x.data_ptr<float>(),
                     centers.data_ptr<float>(), gx.data_ptr<float>(),
                     gcp.data_ptr<float>(), L, (float)sigma, n);
  return {gx, gcp.sum(0)};
}

}  // namespace dsin
