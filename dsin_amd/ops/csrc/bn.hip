// Fused batch-norm (+activation) kernel family, v2.
//
// torch.nn.BatchNorm2d semantics: normalize with the BIASED batch variance,
// running_var updated with the UNBIASED variance,
// running = (1-momentum)*running + momentum*batch. bf16 data, fp32 stats.
//
// Layout-aware: NCHW is channel-major, so every kernel runs a 2D grid
// (spatial chunks x B*C) — the channel index is one divide per BLOCK and
// all loads are 16B-vectorized.
//
//  bn_stats_part -> bn_finalize: per-channel mean/rstd (+EMA, scale/shift)
//  bn_apply:       out = act(scale*y + shift)
//  bn_bwd_part:    per-channel sum(dy_eff), sum(dy_eff*xhat), act' inline
//  bn_bwd_apply:   dx = gamma*rstd*(dy_eff - s1/n - xhat*s2/n)

#include "common.h"

namespace dsin {

using bnbf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) unsigned short bn_u16x8;

__device__ __forceinline__ float bnb2f(unsigned short u) {
  bnbf16 v = *reinterpret_cast<bnbf16*>(&u);
  return __bfloat162float(v);
}

__device__ __forceinline__ void block_reduce2(float& a, float& b) {
  a = wave_reduce_sum(a);
  b = wave_reduce_sum(b);
  __shared__ float red[2][4];
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) {
    red[0][wid] = a;
    red[1][wid] = b;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < (int)(blockDim.x >> 6); ++w) {
      a += red[0][w];
      b += red[1][w];
    }
  }
}

__global__ void bn_stats_part_kernel(const bnbf16* __restrict__ y,
                                     float* __restrict__ psum,   // (S, C)
                                     float* __restrict__ psum2,  // (S, C)
                                     int C, long long HW, int B, int S) {
  const int c = blockIdx.y;
  const int sidx = blockIdx.x;
  float s = 0.f, s2 = 0.f;
  for (int b = 0; b < B; ++b) {
    const bnbf16* p = y + ((long long)b * C + c) * HW;
    long long i0 = (long long)(sidx * (int)blockDim.x + threadIdx.x) * 8;
    long long stride = (long long)S * blockDim.x * 8;
    for (long long i = i0; i + 7 < HW; i += stride) {
      const bn_u16x8 v = *reinterpret_cast<const bn_u16x8*>(&p[i]);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float f = bnb2f(v[k]);
        s += f;
        s2 += f * f;
      }
    }
    if (sidx == 0 && threadIdx.x < (HW & 7)) {  // tail
      float f = __bfloat162float(p[(HW & ~7LL) + threadIdx.x]);
      s += f;
      s2 += f * f;
    }
  }
  block_reduce2(s, s2);
  // per-slice plain stores into (S, C) partials: no atomics, and the
  // buffers need no zero-fill kernel (two fills per BN call was ~0.6 ms
  // per training step); bn_finalize sums the S slices.
  if (threadIdx.x == 0) {
    psum[sidx * C + c] = s;
    psum2[sidx * C + c] = s2;
  }
}

// Applies the normalization AND derives scale/shift in-block from the
// (S, C) partial sums (train) or the running stats (eval): the separate
// bn_finalize launch (one tiny kernel per BN call, ~0.6 ms/step across the
// model) is gone. The first block of each channel also publishes
// mean/rstd for the backward and advances the EMA running stats.
__global__ void bn_apply_kernel(const bnbf16* __restrict__ y,
                                const bnbf16* __restrict__ res,  // or null
                                bnbf16* __restrict__ out,
                                const float* __restrict__ psum,  // (S,C)|null
                                const float* __restrict__ psum2,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                float* __restrict__ mean_out,
                                float* __restrict__ rstd_out,
                                float* __restrict__ rmean,
                                float* __restrict__ rvar,
                                int C, long long HW, int S, float n,
                                float momentum, float eps, int training,
                                int act) {
  const int bc = blockIdx.y;
  const int c = bc % C;
  float m, rs;
  if (training) {
    float t1 = 0.f, t2 = 0.f;
    for (int sdx = 0; sdx < S; ++sdx) {
      t1 += psum[sdx * C + c];
      t2 += psum2[sdx * C + c];
    }
    m = t1 / n;
    const float var = fmaxf(t2 / n - m * m, 0.f);
    rs = rsqrtf(var + eps);
    if (blockIdx.x == 0 && bc < C && threadIdx.x == 0) {
      mean_out[c] = m;
      rstd_out[c] = rs;
      const float unbias = (n > 1.f) ? var * n / (n - 1.f) : var;
      rmean[c] = (1.f - momentum) * rmean[c] + momentum * m;
      rvar[c] = (1.f - momentum) * rvar[c] + momentum * unbias;
    }
  } else {
    m = rmean[c];
    rs = rsqrtf(rvar[c] + eps);
    if (blockIdx.x == 0 && bc < C && threadIdx.x == 0) {
      mean_out[c] = m;
      rstd_out[c] = rs;
    }
  }
  const float sc = gamma[c] * rs, sh = beta[c] - m * sc;
  const bnbf16* p = y + (long long)bc * HW;
  const bnbf16* q = res ? res + (long long)bc * HW : nullptr;
  bnbf16* o = out + (long long)bc * HW;
  long long i0 = (long long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long i = i0; i + 7 < HW; i += stride) {
    const bn_u16x8 v = *reinterpret_cast<const bn_u16x8*>(&p[i]);
    bn_u16x8 rv;
    if (q) rv = *reinterpret_cast<const bn_u16x8*>(&q[i]);
    bn_u16x8 r;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = bnb2f(v[k]) * sc + sh;
      if (q) f += bnb2f(rv[k]);
      if (act == 1) f = fmaxf(f, 0.f);
      else if (act == 2) f = fmaxf(f, 0.2f * f);
      bnbf16 h = __float2bfloat16(f);
      r[k] = *reinterpret_cast<unsigned short*>(&h);
    }
    *reinterpret_cast<bn_u16x8*>(&o[i]) = r;
  }
  if (blockIdx.x == 0 && threadIdx.x < (HW & 7)) {
    long long j = (HW & ~7LL) + threadIdx.x;
    float f = __bfloat162float(p[j]) * sc + sh;
    if (q) f += __bfloat162float(q[j]);
    if (act == 1) f = fmaxf(f, 0.f);
    else if (act == 2) f = fmaxf(f, 0.2f * f);
    o[j] = __float2bfloat16(f);
  }
}

__device__ __forceinline__ float bn_actp(float g, float ov, int act) {
  if (act == 1) return ov > 0.f ? g : 0.f;
  if (act == 2) return ov > 0.f ? g : 0.2f * g;
  return g;
}

__global__ void bn_bwd_part_kernel(const bnbf16* __restrict__ dy,
                                   const bnbf16* __restrict__ y,
                                   const bnbf16* __restrict__ out,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   float* __restrict__ s1,
                                   float* __restrict__ s2,
                                   int C, long long HW, int B, int S,
                                   int act) {
  const int c = blockIdx.y;
  const int sidx = blockIdx.x;
  const float m = mean[c], rs = rstd[c];
  float a = 0.f, b2 = 0.f;
  for (int b = 0; b < B; ++b) {
    const long long base = ((long long)b * C + c) * HW;
    long long i0 = (long long)(sidx * (int)blockDim.x + threadIdx.x) * 8;
    long long stride = (long long)S * blockDim.x * 8;
    for (long long i = i0; i + 7 < HW; i += stride) {
      const bn_u16x8 gv = *reinterpret_cast<const bn_u16x8*>(&dy[base + i]);
      const bn_u16x8 yv = *reinterpret_cast<const bn_u16x8*>(&y[base + i]);
      const bn_u16x8 ov = *reinterpret_cast<const bn_u16x8*>(&out[base + i]);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = bn_actp(bnb2f(gv[k]), bnb2f(ov[k]), act);
        a += g;
        b2 += g * (bnb2f(yv[k]) - m) * rs;
      }
    }
    if (sidx == 0 && threadIdx.x < (HW & 7)) {
      long long j = base + (HW & ~7LL) + threadIdx.x;
      float g = bn_actp(__bfloat162float(dy[j]), __bfloat162float(out[j]), act);
      a += g;
      b2 += g * (__bfloat162float(y[j]) - m) * rs;
    }
  }
  block_reduce2(a, b2);
  // plain per-slice stores (deterministic: the host reduces the (S, C)
  // partials with one ordered at::sum — fp32 atomicAdd order varies
  // between runs and was the one nondeterminism in the BN backward)
  if (threadIdx.x == 0) {
    s1[(long long)sidx * C + c] = a;
    s2[(long long)sidx * C + c] = b2;
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
                                    int C, long long HW, float invn, int act,
                                    int training) {
  const int bc = blockIdx.y;
  const int c = bc % C;
  const float m = mean[c], rs = rstd[c], gm = gamma[c];
  const float t1 = s1[c] * invn, t2 = s2[c] * invn;
  const long long base = (long long)bc * HW;
  long long i0 = (long long)(blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (long long i = i0; i + 7 < HW; i += stride) {
    const bn_u16x8 gv = *reinterpret_cast<const bn_u16x8*>(&dy[base + i]);
    const bn_u16x8 ov = *reinterpret_cast<const bn_u16x8*>(&out[base + i]);
    const bn_u16x8 yv = *reinterpret_cast<const bn_u16x8*>(&y[base + i]);
    bn_u16x8 r;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float g = bn_actp(bnb2f(gv[k]), bnb2f(ov[k]), act);
      float v;
      if (training) {
        const float xh = (bnb2f(yv[k]) - m) * rs;
        v = gm * rs * (g - t1 - xh * t2);
      } else {
        v = gm * rs * g;
      }
      bnbf16 h = __float2bfloat16(v);
      r[k] = *reinterpret_cast<unsigned short*>(&h);
    }
    *reinterpret_cast<bn_u16x8*>(&dx[base + i]) = r;
  }
  if (blockIdx.x == 0 && threadIdx.x < (HW & 7)) {
    long long j = base + (HW & ~7LL) + threadIdx.x;
    float g = bn_actp(__bfloat162float(dy[j]), __bfloat162float(out[j]), act);
    float v;
    if (training) {
      const float xh = (__bfloat162float(y[j]) - m) * rs;
      v = gm * rs * (g - t1 - xh * t2);
    } else {
      v = gm * rs * g;
    }
    dx[j] = __float2bfloat16(v);
  }
}

// --------------------------------------------------------------- host

static int _spatial_chunks(long long HW) {
  return (int)std::min<long long>(std::max<long long>(HW / (256 * 8), 1), 32);
}

std::vector<torch::Tensor> bn_fwd(torch::Tensor y, torch::Tensor gamma,
                                  torch::Tensor beta, torch::Tensor rmean,
                                  torch::Tensor rvar, double momentum,
                                  double eps, bool training, int64_t act,
                                  c10::optional<torch::Tensor> residual) {
  CHECK_CUDA_CONTIG(y);
  TORCH_CHECK(y.scalar_type() == torch::kBFloat16, "bn: y must be bf16");
  const int B = (int)y.size(0), C = (int)y.size(1);
  const long long HW = (long long)y.size(2) * y.size(3);
  auto optsF = y.options().dtype(torch::kFloat32);
  auto mean = torch::empty({C}, optsF);
  auto rstd = torch::empty({C}, optsF);
  auto stream = at::cuda::getCurrentCUDAStream();
  const int S = _spatial_chunks(HW);
  float* psum = nullptr;
  float* psum2 = nullptr;
  torch::Tensor parts;
  if (training) {
    parts = torch::empty({2, S, C}, optsF);  // written fully: no fill
    psum = parts.data_ptr<float>();
    psum2 = psum + (long long)S * C;
    hipLaunchKernelGGL(bn_stats_part_kernel, dim3(S, C), dim3(256), 0, stream,
                       (const bnbf16*)y.data_ptr(), psum, psum2, C, HW, B, S);
  }
  auto out = torch::empty_like(y);
  const bnbf16* resp = nullptr;
  if (residual.has_value()) {
    CHECK_CUDA_CONTIG(residual.value());
    resp = (const bnbf16*)residual->data_ptr();
  }
  hipLaunchKernelGGL(bn_apply_kernel,
                     dim3(S, B * C), dim3(256), 0, stream,
                     (const bnbf16*)y.data_ptr(), resp, (bnbf16*)out.data_ptr(),
                     psum, psum2, gamma.data_ptr<float>(),
                     beta.data_ptr<float>(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), rmean.data_ptr<float>(),
                     rvar.data_ptr<float>(), C, HW, S,
                     (float)((long long)B * HW), (float)momentum, (float)eps,
                     training ? 1 : 0, (int)act);
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
  auto stream = at::cuda::getCurrentCUDAStream();
  const int S = _spatial_chunks(HW);
  // (2, S, C) plain-store partials -> one ordered sum (deterministic)
  auto parts = torch::empty({2, S, C}, optsF);
  hipLaunchKernelGGL(bn_bwd_part_kernel, dim3(S, C), dim3(256), 0, stream,
                     (const bnbf16*)dy.data_ptr(), (const bnbf16*)y.data_ptr(),
                     (const bnbf16*)out.data_ptr(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), parts[0].data_ptr<float>(),
                     parts[1].data_ptr<float>(), C, HW, B, S, (int)act);
  auto sbuf = parts.sum(1);  // (2, C), contiguous
  auto s1 = sbuf[0];
  auto s2 = sbuf[1];
  auto dx = torch::empty_like(y);
  hipLaunchKernelGGL(bn_bwd_apply_kernel, dim3(_spatial_chunks(HW), B * C),
                     dim3(256), 0, stream, (const bnbf16*)dy.data_ptr(),
                     (const bnbf16*)y.data_ptr(), (const bnbf16*)out.data_ptr(),
                     (bnbf16*)dx.data_ptr(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), gamma.data_ptr<float>(),
                     s1.data_ptr<float>(), s2.data_ptr<float>(), C, HW,
                     1.f / (float)(B * HW), (int)act, training ? 1 : 0);
  // dgamma = s2, dbeta = s1
  return {dx, s2, s1};
}

}  // namespace dsin
