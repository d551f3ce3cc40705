// Verify hypothesized gfx950 mfma_f32_16x16x32_bf16 fragment layouts with a
// random asymmetric GEMM vs CPU (CDNA4 guide G9 rule):
//   A slot: lane, reg -> A[row = lane&15][k = 8*(lane>>4) + reg]
//   B slot: lane, reg -> B[k = 8*(lane>>4) + reg][col = lane&15]
//   C slot: lane, reg -> C[row = (lane>>4)*4 + reg][col = lane&15]
// Build: hipcc --offload-arch=gfx950 -O3 scripts/mfma_verify.hip -o /tmp/mfma_verify

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__global__ void gemm16(const unsigned short* A, const unsigned short* B, float* C) {
  const int lane = threadIdx.x;
  bf16x8 a, b;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    a[i] = (short)A[(lane & 15) * 32 + 8 * (lane >> 4) + i];
    b[i] = (short)B[(8 * (lane >> 4) + i) * 16 + (lane & 15)];
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) C[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = c[r];
}

static unsigned short f2bf_h(float f) {
  unsigned int i;
  __builtin_memcpy(&i, &f, 4);
  unsigned int lsb = (i >> 16) & 1;
  return (unsigned short)((i + 0x7FFFu + lsb) >> 16);
}

static float bf2f_h(unsigned short u) {
  unsigned int i = ((unsigned int)u) << 16;
  float f;
  __builtin_memcpy(&f, &i, 4);
  return f;
}

int main() {
  srand(7);
  unsigned short A[16 * 32], B[32 * 16];
  for (int i = 0; i < 16 * 32; ++i) A[i] = f2bf_h(((rand() % 2000) - 1000) / 500.0f);
  for (int i = 0; i < 32 * 16; ++i) B[i] = f2bf_h(((rand() % 2000) - 1000) / 500.0f);
  unsigned short *dA, *dB;
  float* dC;
  hipMalloc(&dA, sizeof(A));
  hipMalloc(&dB, sizeof(B));
  hipMalloc(&dC, 16 * 16 * 4);
  hipMemcpy(dA, A, sizeof(A), hipMemcpyHostToDevice);
  hipMemcpy(dB, B, sizeof(B), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(gemm16, dim3(1), dim3(64), 0, 0, dA, dB, dC);
  hipDeviceSynchronize();
  float C[256];
  hipMemcpy(C, dC, sizeof(C), hipMemcpyDeviceToHost);
  float max_err = 0.f;
  for (int r = 0; r < 16; ++r)
    for (int n = 0; n < 16; ++n) {
      float ref = 0.f;
      for (int k = 0; k < 32; ++k) ref += bf2f_h(A[r * 32 + k]) * bf2f_h(B[k * 16 + n]);
      max_err = fmaxf(max_err, fabsf(ref - C[r * 16 + n]));
    }
  printf("mfma 16x16x32 bf16 layout verify: max_err=%g -> %s\n", max_err,
         max_err < 1e-3f ? "LAYOUTS CONFIRMED" : "LAYOUT WRONG");
  return max_err < 1e-3f ? 0 : 1;
}
