#include "hip/hip_runtime.h"
// Fused softmax-cross-entropy bitcost over the L centers dimension.
//
// Mirror of /root/reference/src/probclass_imgcomp.py:100-106:
//   bits[n,c,h,w] = CE(logits[n,:,c,h,w], symbols[n,c,h,w]) * log2(e)
// logits layout (N, L, S) with S = C*H*W flattened spatial+channel sites.
// Backward: dlogits = g * log2e * (softmax(logits) - onehot(sym)).

#include "common_hip.h"

namespace dsin {

constexpr int MAX_LOGITS = 16;
constexpr float LOG2E = 1.4426950408889634f;

__global__ void bitcost_ce_fwd_kernel(const float* __restrict__ logits,
                                      const int64_t* __restrict__ symbols,
                                      float* __restrict__ bits,
                                      int L, int64_t S, int64_t total) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int64_t n = i / S, s = i % S;
    const float* lg = logits + (n * L) * S + s;
    float m = -1e30f;
#pragma unroll 8
    for (int l = 0; l < L; ++l) m = fmaxf(m, lg[l * S]);
    float denom = 0.f;
#pragma unroll 8
    for (int l = 0; l < L; ++l) denom += __expf(lg[l * S] - m);
    int sym = (int)symbols[i];
    // CE = logsumexp - logit[sym]
    bits[i] = (logf(denom) + m - lg[sym * S]) * LOG2E;
  }
}

__global__ void bitcost_ce_bwd_kernel(const float* __restrict__ g,
                                      const float* __restrict__ logits,
                                      const int64_t* __restrict__ symbols,
                                      float* __restrict__ glogits,
                                      int L, int64_t S, int64_t total) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int64_t n = i / S, s = i % S;
    const float* lg = logits + (n * L) * S + s;
    float* gl = glogits + (n * L) * S + s;
    float m = -1e30f;
#pragma unroll 8
    for (int l = 0; l < L; ++l) m = fmaxf(m, lg[l * S]);
    float denom = 0.f;
    float e[MAX_LOGITS];
#pragma unroll 8
    for (int l = 0; l < L; ++l) { e[l] = __expf(lg[l * S] - m); denom += e[l]; }
    float inv = 1.f / denom;
    int sym = (int)symbols[i];
    float gi = g[i] * LOG2E;
#pragma unroll 8
    for (int l = 0; l < L; ++l)
      gl[l * S] = gi * (e[l] * inv - (l == sym ? 1.f : 0.f));
  }
}

torch::Tensor bitcost_ce_fwd(torch::Tensor logits, torch::Tensor symbols) {
  CHECK_CUDA_CONTIG(logits);
  CHECK_CUDA_CONTIG(symbols);
  TORCH_CHECK(logits.dim() == 5, "logits must be (N, L, C, H, W)");
  TORCH_CHECK(logits.scalar_type() == torch::kFloat32, "logits must be fp32");
  int64_t N = logits.size(0), L = logits.size(1);
  TORCH_CHECK(L <= MAX_LOGITS, "at most ", MAX_LOGITS, " logit classes");
  int64_t S = logits.numel() / (N * L);
  TORCH_CHECK(symbols.numel() == N * S, "symbols shape mismatch");
  auto bits = torch::empty(symbols.sizes(), logits.options());
  int64_t total = N * S;
  int block = 256;
  int grid = std::min<int64_t>((total + block - 1) / block, 4096);
  hipLaunchKernelGGL(bitcost_ce_fwd_kernel, dim3(grid), dim3(block), 0,
                     at::hip::getCurrentHIPStreamMasqueradingAsCUDA(),
                     logits.data_ptr<float>(), symbols.data_ptr<int64_t>(),
                     bits.data_ptr<float>(), (int)L, S, total);
  return bits;
}

torch::Tensor bitcost_ce_bwd(torch::Tensor g, torch::Tensor logits,
                             torch::Tensor symbols) {
  CHECK_CUDA_CONTIG(g);
  CHECK_CUDA_CONTIG(logits);
  CHECK_CUDA_CONTIG(symbols);
  int64_t N = logits.size(0), L = logits.size(1);
  int64_t S = logits.numel() / (N * L);
  auto glogits = torch::empty_like(logits);
  int64_t total = N * S;
  int block = 256;
  int grid = std::min<int64_t>((total + block - 1) / block, 4096);
  hipLaunchKernelGGL(bitcost_ce_bwd_kernel, dim3(grid), dim3(block), 0,
                     at::hip::getCurrentHIPStreamMasqueradingAsCUDA(),
                     g.data_ptr<float>(), logits.data_ptr<float>(),
                     symbols.data_ptr<int64_t>(), glogits.data_ptr<float>(),
                     (int)L, S, total);
  return glogits;
}

}  // namespace dsin
