#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
typedef __attribute__((ext_vector_type(8))) __bf16 b8;
typedef __attribute__((ext_vector_type(16))) float f16v;

__global__ void k_const(float* C) {
  int lane = threadIdx.x & 63;
  for (int reg = 0; reg < 16; ++reg) {
    int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
    C[row * 32 + (lane & 31)] = 1.0f;
  }
}
__global__ void k_loads(const float* A, const float* B, float* C) {
  int lane = threadIdx.x & 63;
  float s = 0.f;
  for (int e = 0; e < 8; ++e) {
    int k = (lane >> 5) * 8 + e;
    s += A[(lane & 31) * 16 + k] + B[k * 32 + (lane & 31)];
  }
  for (int reg = 0; reg < 16; ++reg) {
    int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
    C[row * 32 + (lane & 31)] = s;
  }
}
__global__ void k_full(const float* A, const float* B, float* C) {
  int lane = threadIdx.x & 63;
  b8 a, b;
  for (int e = 0; e < 8; ++e) {
    int k = (lane >> 5) * 8 + e;
    a[e] = (__bf16)A[(lane & 31) * 16 + k];
    b[e] = (__bf16)B[k * 32 + (lane & 31)];
  }
  f16v acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  for (int reg = 0; reg < 16; ++reg) {
    int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
    C[row * 32 + (lane & 31)] = acc[reg];
  }
}
int main() {
  float *A, *B, *C;
  hipMalloc(&A, 512 * 4); hipMalloc(&B, 512 * 4); hipMalloc(&C, 1024 * 4);
  float hA[512], hB[512], hC[1024];
  for (int i = 0; i < 512; ++i) { hA[i] = (i % 7) * 0.25f - 0.5f; hB[i] = ((i * 3) % 11) * 0.125f - 0.4f; }
  hipMemcpy(A, hA, 2048, hipMemcpyHostToDevice);
  hipMemcpy(B, hB, 2048, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(k_const, dim3(1), dim3(64), 0, 0, C);
  printf("const: %s\n", hipGetErrorString(hipDeviceSynchronize()));
  hipLaunchKernelGGL(k_loads, dim3(1), dim3(64), 0, 0, A, B, C);
  printf("loads: %s\n", hipGetErrorString(hipDeviceSynchronize()));
  hipLaunchKernelGGL(k_full, dim3(1), dim3(64), 0, 0, A, B, C);
  printf("full: %s\n", hipGetErrorString(hipDeviceSynchronize()));
  hipMemcpy(hC, C, 4096, hipMemcpyDeviceToHost);
  // check vs reference
  double maxerr = 0;
  for (int r = 0; r < 32; ++r)
    for (int c = 0; c < 32; ++c) {
      double e = 0;
      for (int k = 0; k < 16; ++k) e += (double)(__bf16)hA[r*16+k] * (double)(__bf16)hB[k*32+c];
      double d = fabs(e - hC[r*32+c]);
      if (d > maxerr) maxerr = d;
    }
  printf("maxerr %f\n", maxerr);
  return 0;
}
