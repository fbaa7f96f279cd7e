// Shared helpers for the dsin_amd CDNA4 (gfx950) kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#define DSIN_CHECK_HIP(expr)                                                   \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));       \
  } while (0)

#define CHECK_CUDA_CONTIG(t)                                                   \
  TORCH_CHECK((t).is_cuda() && (t).is_contiguous(), #t " must be contiguous on GPU")

namespace dsin {

using bf16 = __hip_bfloat16;

__device__ __forceinline__ float b2f(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2b(float v) { return __float2bfloat16(v); }

constexpr int WAVE = 64;

// order-preserving encode of a float into uint32 (for packed atomic argmax)
__device__ __forceinline__ unsigned int float_flip(float f) {
  unsigned int u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

inline dim3 grid1d(int64_t n, int block) {
  return dim3(static_cast<unsigned int>((n + block - 1) / block));
}

}  // namespace dsin
