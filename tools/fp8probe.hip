#include <hip/hip_runtime.h>
#include <hip/hip_fp8.h>
#include <cstdio>
typedef __attribute__((ext_vector_type(4))) float f4;
typedef long long i64;

__global__ void k_full(const float* A, const float* B, float* C) {
  int lane = threadIdx.x & 63;
  unsigned char ab[8], bb[8];
  for (int e = 0; e < 8; ++e) {
    int k = (lane >> 4) * 8 + e;
    __hip_fp8_e4m3 va(A[(lane & 15) * 32 + k]);
    __hip_fp8_e4m3 vb(B[k * 16 + (lane & 15)]);
    ab[e] = va.__x;
    bb[e] = vb.__x;
  }
  i64 a = *reinterpret_cast<i64*>(ab);
  i64 b = *reinterpret_cast<i64*>(bb);
  f4 acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a, b, acc, 0, 0, 0);
  for (int reg = 0; reg < 4; ++reg) {
    int row = (lane >> 4) * 4 + reg, col = lane & 15;
    C[row * 16 + col] = acc[reg];
  }
}
int main() {
  float *A, *B, *C;
  hipMalloc(&A, 512*4); hipMalloc(&B, 512*4); hipMalloc(&C, 256*4);
  float hA[512], hB[512], hC[256];
  for (int i = 0; i < 512; ++i) { hA[i] = (i%5)*0.25f - 0.5f; hB[i] = ((i*7)%9)*0.25f - 1.0f; }
  hipMemcpy(A,hA,2048,hipMemcpyHostToDevice); hipMemcpy(B,hB,2048,hipMemcpyHostToDevice);
  hipLaunchKernelGGL(k_full, dim3(1), dim3(64), 0, 0, A, B, C);
  printf("launch: %s\n", hipGetErrorString(hipDeviceSynchronize()));
  hipMemcpy(hC,C,1024,hipMemcpyDeviceToHost);
  double maxerr=0;
  for (int r=0;r<16;++r) for(int c=0;c<16;++c){
    double e=0; for(int k=0;k<32;++k){
      // quantize through fp8 on host via float roundtrip approximation:
      e += (double)hA[r*32+k] * (double)hB[k*16+c];  // values are exact in fp8
    }
    double d = fabs(e - hC[r*16+c]); if (d>maxerr) maxerr=d;
  }
  printf("maxerr %f\n", maxerr);
  return 0;
}
