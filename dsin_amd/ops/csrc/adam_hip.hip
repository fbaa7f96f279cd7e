#include "hip/hip_runtime.h"
// Fused flat-buffer Adam step.
//
// TF-semantics Adam (the reference uses tf.train.AdamOptimizer,
// src/training_helpers_imgcomp.py:38-48):
//   g' = g + wd * p    (optional elementwise L2 term: the model's
//                        factor/2*sum(w^2) regularizers enter Adam exactly
//                        as lambda*w gradients, so they are folded into the
//                        step instead of being built in the autograd graph)
//   m = b1 m + (1-b1) g' ;  v = b2 v + (1-b2) g'^2
//   lr_t = lr * sqrt(1 - b2^t) / (1 - b1^t)
//   p  -= lr_t * m / (sqrt(v) + eps)
// One launch per parameter group over flat fp32 buffers (parameters are
// repointed to views of the flat buffer — ops/adam.py). lr and t live in
// device memory so the kernel replays correctly inside hipGraphs while the
// staircase schedule and step count advance.

#include "common_hip.h"

namespace dsin {

__global__ void adam_step_kernel(float* __restrict__ p,
                                 const float* __restrict__ g,
                                 float* __restrict__ m,
                                 float* __restrict__ v,
                                 const float* __restrict__ lr,
                                 const int* __restrict__ step,
                                 const float* __restrict__ wd,
                                 float b1, float b2, float eps, long long n) {
  const float t = (float)*step;
  const float corr = lr[0] * sqrtf(1.f - powf(b2, t)) / (1.f - powf(b1, t));
  long long i = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long long stride = (long long)gridDim.x * blockDim.x * 4;
  for (; i + 3 < n; i += stride) {
    float4 gp = *reinterpret_cast<const float4*>(&g[i]);
    float4 mp = *reinterpret_cast<float4*>(&m[i]);
    float4 vp = *reinterpret_cast<float4*>(&v[i]);
    float4 pp = *reinterpret_cast<float4*>(&p[i]);
    float* gf = &gp.x;
    float* mf = &mp.x;
    float* vf = &vp.x;
    float* pf = &pp.x;
    if (wd != nullptr) {
      const float4 wp = *reinterpret_cast<const float4*>(&wd[i]);
      const float* wf = &wp.x;
#pragma unroll
      for (int k = 0; k < 4; ++k) gf[k] += wf[k] * pf[k];
    }
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      mf[k] = b1 * mf[k] + (1.f - b1) * gf[k];
      vf[k] = b2 * vf[k] + (1.f - b2) * gf[k] * gf[k];
      pf[k] -= corr * mf[k] / (sqrtf(vf[k]) + eps);
    }
    *reinterpret_cast<float4*>(&m[i]) = mp;
    *reinterpret_cast<float4*>(&v[i]) = vp;
    *reinterpret_cast<float4*>(&p[i]) = pp;
  }
  // tail (n % 4)
  if (blockIdx.x == 0 && threadIdx.x < 4) {
    long long base = n & ~3LL;
    long long j = base + threadIdx.x;
    if (j < n) {
      float gv = g[j] + (wd != nullptr ? wd[j] * p[j] : 0.f);
      float mv = b1 * m[j] + (1.f - b1) * gv;
      float vv = b2 * v[j] + (1.f - b2) * gv * gv;
      m[j] = mv;
      v[j] = vv;
      p[j] -= corr * mv / (sqrtf(vv) + eps);
    }
  }
}

void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, torch::Tensor lr, torch::Tensor step,
               c10::optional<torch::Tensor> wd, double b1, double b2,
               double eps) {
  CHECK_CUDA_CONTIG(p);
  CHECK_CUDA_CONTIG(g);
  CHECK_CUDA_CONTIG(m);
  CHECK_CUDA_CONTIG(v);
  int64_t n = p.numel();
  int grid = (int)std::min<int64_t>((n / 4 + 255) / 256, 2048);
  const float* wdp = nullptr;
  if (wd.has_value()) {
    CHECK_CUDA_CONTIG(wd.value());
    wdp = wd->data_ptr<float>();
  }
  hipLaunchKernelGGL(adam_step_kernel, dim3(std::max(grid, 1)), dim3(256), 0,
                     at::hip::getCurrentHIPStreamMasqueradingAsCUDA(), p.data_ptr<float>(),
                     g.data_ptr<float>(), m.data_ptr<float>(),
                     v.data_ptr<float>(), lr.data_ptr<float>(),
                     step.data_ptr<int>(), wdp, (float)b1, (float)b2,
                     (float)eps, n);
}

}  // namespace dsin
