#include "hip/hip_runtime.h"
// Fused batch-norm (+activation) kernel family.
//
// Replaces MIOpen's separate BN-forward / BN-backward / activation kernels
// for the conv stack. torch.nn.BatchNorm2d semantics: normalize with the
// BIASED batch variance, running_var updated with the UNBIASED variance,
// running = (1-momentum)*running + momentum*batch. bf16 data, fp32 stats.
//
//  bn_stats:  per-channel mean/rstd (+EMA update, +precomputed scale/shift)
//  bn_apply:  out = act(scale*y + shift)      [one elementwise pass]
//  bn_bwd_reduce: per-channel sum(dy_eff), sum(dy_eff * xhat) with the
//                 activation derivative applied inline from the saved output
//  bn_bwd_apply:  dx = gamma*rstd*(dy_eff - s1/n - xhat * s2/n); also emits
//                 dgamma = s2, dbeta = s1

#include "common_hip.h"

namespace dsin {

using bnbf16 = __hip_bfloat16;

__global__ void bn_stats_kernel(const bnbf16* __restrict__ y,
                                float* __restrict__ mean,
                                float* __restrict__ rstd,
                                float* __restrict__ scale,
                                float* __restrict__ shift,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                float* __restrict__ rmean,
                                float* __restrict__ rvar,
                                int C, long long HW, int B, long long cstride,
                                float momentum, float eps, int training) {
  const int c = blockIdx.x;
  float s = 0.f, s2 = 0.f;
  if (training) {
    for (int b = 0; b < B; ++b) {
      const bnbf16* p = y + b * cstride * C + c * cstride;
      for (long long i = threadIdx.x; i < HW; i += blockDim.x) {
        float v = __bfloat162float(p[i]);
        s += v;
        s2 += v * v;
      }
    }
    s = wave_reduce_sum(s);
    s2 = wave_reduce_sum(s2);
    __shared__ float red[2][4];
    const int wid = threadIdx.x >> 6;
    if ((threadIdx.x & 63) == 0) {
      red[0][wid] = s;
      red[1][wid] = s2;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      float ts = 0.f, ts2 = 0.f;
      for (int w = 0; w < (int)(blockDim.x >> 6); ++w) {
        ts += red[0][w];
        ts2 += red[1][w];
      }
      const float n = (float)(HW * B);
      const float m = ts / n;
      float var = ts2 / n - m * m;
      var = fmaxf(var, 0.f);
      const float rs = rsqrtf(var + eps);
      mean[c] = m;
      rstd[c] = rs;
      // torch: running_var uses the unbiased estimate
      const float unbias = (n > 1.f) ? var * n / (n - 1.f) : var;
      rmean[c] = (1.f - momentum) * rmean[c] + momentum * m;
      rvar[c] = (1.f - momentum) * rvar[c] + momentum * unbias;
      const float sc = gamma[c] * rs;
      scale[c] = sc;
      shift[c] = beta[c] - m * sc;
    }
  } else if (threadIdx.x == 0) {
    const float m = rmean[c];
    const float rs = rsqrtf(rvar[c] + eps);
    mean[c] = m;
    rstd[c] = rs;
    const float sc = gamma[c] * rs;
    scale[c] = sc;
    shift[c] = beta[c] - m * sc;
  }
}

__global__ void bn_apply_kernel(const bnbf16* __restrict__ y,
                                bnbf16* __restrict__ out,
                                const float* __restrict__ scale,
                                const float* __restrict__ shift,
                                int C, long long HW, long long total,
                                int act) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int c = (int)((i / HW) % C);
    float v = __bfloat162float(y[i]) * scale[c] + shift[c];
    if (act == 1) v = fmaxf(v, 0.f);
    else if (act == 2) v = fmaxf(v, 0.2f * v);
    out[i] = __float2bfloat16(v);
  }
}

__global__ void bn_bwd_reduce_kernel(const bnbf16* __restrict__ dy,
                                     const bnbf16* __restrict__ y,
                                     const bnbf16* __restrict__ out,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     float* __restrict__ s1,  // (C,) sum dy
                                     float* __restrict__ s2,  // (C,) sum dy*xh
                                     int C, long long HW, int B,
                                     long long cstride, int act) {
  const int c = blockIdx.x;
  const float m = mean[c], rs = rstd[c];
  float a = 0.f, b2 = 0.f;
  for (int b = 0; b < B; ++b) {
    const long long base = b * cstride * C + c * cstride;
    for (long long i = threadIdx.x; i < HW; i += blockDim.x) {
      float g = __bfloat162float(dy[base + i]);
      if (act == 1) g = (__bfloat162float(out[base + i]) > 0.f) ? g : 0.f;
      else if (act == 2)
        g = (__bfloat162float(out[base + i]) > 0.f) ? g : 0.2f * g;
      const float xh = (__bfloat162float(y[base + i]) - m) * rs;
      a += g;
      b2 += g * xh;
    }
  }
  a = wave_reduce_sum(a);
  b2 = wave_reduce_sum(b2);
  __shared__ float red[2][4];
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) {
    red[0][wid] = a;
    red[1][wid] = b2;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float ta = 0.f, tb = 0.f;
    for (int w = 0; w < (int)(blockDim.x >> 6); ++w) {
      ta += red[0][w];
      tb += red[1][w];
    }
    s1[c] = ta;
    s2[c] = tb;
  }
}

__global__ void bn_bwd_apply_kernel(const bnbf16* __restrict__ dy,
                                    const bnbf16* __restrict__ y,
                                    const bnbf16* __restrict__ out,
                                    bnbf16* __restrict__ dx,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ s1,
                                    const float* __restrict__ s2,
                                    int C, long long HW, long long total,
                                    float invn, int act, int training) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int c = (int)((i / HW) % C);
    float g = __bfloat162float(dy[i]);
    if (act == 1) g = (__bfloat162float(out[i]) > 0.f) ? g : 0.f;
    else if (act == 2) g = (__bfloat162float(out[i]) > 0.f) ? g : 0.2f * g;
    float v;
    if (training) {
      const float xh = (__bfloat162float(y[i]) - mean[c]) * rstd[c];
      v = gamma[c] * rstd[c] * (g - s1[c] * invn - xh * (s2[c] * invn));
    } else {
      v = gamma[c] * rstd[c] * g;  // eval: stats are constants
    }
    dx[i] = __float2bfloat16(v);
  }
}

// --------------------------------------------------------------- host

std::vector<torch::Tensor> bn_fwd(torch::Tensor y, torch::Tensor gamma,
                                  torch::Tensor beta, torch::Tensor rmean,
                                  torch::Tensor rvar, double momentum,
                                  double eps, bool training, int64_t act) {
  CHECK_CUDA_CONTIG(y);
  TORCH_CHECK(y.scalar_type() == torch::kBFloat16, "bn: y must be bf16");
  const int B = (int)y.size(0), C = (int)y.size(1);
  const long long HW = (long long)y.size(2) * y.size(3);
  auto optsF = y.options().dtype(torch::kFloat32);
  auto mean = torch::empty({C}, optsF);
  auto rstd = torch::empty({C}, optsF);
  auto scale = torch::empty({C}, optsF);
  auto shift = torch::empty({C}, optsF);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  hipLaunchKernelGGL(bn_stats_kernel, dim3(C), dim3(256), 0, stream,
                     (const bnbf16*)y.data_ptr(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), scale.data_ptr<float>(),
                     shift.data_ptr<float>(), gamma.data_ptr<float>(),
                     beta.data_ptr<float>(), rmean.data_ptr<float>(),
                     rvar.data_ptr<float>(), C, HW, B, HW,
                     (float)momentum, (float)eps, training ? 1 : 0);
  auto out = torch::empty_like(y);
  long long total = (long long)B * C * HW;
  int grid = (int)std::min<long long>((total + 255) / 256, 8192);
  hipLaunchKernelGGL(bn_apply_kernel, dim3(grid), dim3(256), 0, stream,
                     (const bnbf16*)y.data_ptr(), (bnbf16*)out.data_ptr(),
                     scale.data_ptr<float>(), shift.data_ptr<float>(), C, HW,
                     total, (int)act);
  return {out, mean, rstd};
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor y,
                                  torch::Tensor out, torch::Tensor mean,
                                  torch::Tensor rstd, torch::Tensor gamma,
                                  bool training, int64_t act) {
  CHECK_CUDA_CONTIG(dy);
  CHECK_CUDA_CONTIG(y);
  const int B = (int)y.size(0), C = (int)y.size(1);
  const long long HW = (long long)y.size(2) * y.size(3);
  auto optsF = y.options().dtype(torch::kFloat32);
  auto s1 = torch::empty({C}, optsF);
  auto s2 = torch::empty({C}, optsF);
  auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
  hipLaunchKernelGGL(bn_bwd_reduce_kernel, dim3(C), dim3(256), 0, stream,
                     (const bnbf16*)dy.data_ptr(), (const bnbf16*)y.data_ptr(),
                     (const bnbf16*)out.data_ptr(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), s1.data_ptr<float>(),
                     s2.data_ptr<float>(), C, HW, B, HW, (int)act);
  auto dx = torch::empty_like(y);
  long long total = (long long)B * C * HW;
  int grid = (int)std::min<long long>((total + 255) / 256, 8192);
  hipLaunchKernelGGL(bn_bwd_apply_kernel, dim3(grid), dim3(256), 0, stream,
                     (const bnbf16*)dy.data_ptr(), (const bnbf16*)y.data_ptr(),
                     (const bnbf16*)out.data_ptr(), (bnbf16*)dx.data_ptr(),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     gamma.data_ptr<float>(), s1.data_ptr<float>(),
                     s2.data_ptr<float>(), C, HW, total,
                     1.f / (float)(B * HW), (int)act, training ? 1 : 0);
  // dgamma = s2, dbeta = s1
  return {dx, s2, s1};
}

}  // namespace dsin
