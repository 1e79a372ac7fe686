// Verify mfma_f32_32x32x16_bf16 fragment layouts on gfx950 (asymmetric
// inputs per guide G9).  Assumed layouts under test:
//   A[m][k]: lane holds A[l&31][(l>>5)*8 + j], j=0..7
//   B[k][n]: lane holds B[(l>>5)*8 + j][l&31]
//   C[m][n]: lane holds C[(reg&3) + 8*(reg>>2) + 4*(l>>5)][l&31], reg=0..15
// Also probes permlane32_swap semantics.
//   hipcc --offload-arch=gfx950 -O2 -w scripts/probe_mfma32.cpp -o out && ./out
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

__device__ short f2bf(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  unsigned int rb = 0x7FFF + ((c.i >> 16) & 1);
  return (short)((c.i + rb) >> 16);
}
__device__ float bf2f(short u) {
  union { float f; unsigned int i; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}

__global__ void probe(const float* A, const float* B, float* C) {
  int l = threadIdx.x;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    a[j] = f2bf(A[(l & 31) * 16 + ((l >> 5) * 8 + j)]);
    b[j] = f2bf(B[((l >> 5) * 8 + j) * 32 + (l & 31)]);
  }
  f32x16 acc;
  for (int i = 0; i < 16; ++i) acc[i] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
  for (int reg = 0; reg < 16; ++reg) {
    int row = (reg & 3) + 8 * (reg >> 2) + 4 * (l >> 5);
    C[row * 32 + (l & 31)] = acc[reg];
  }
}

__global__ void probe_permlane(int* out) {
  int l = threadIdx.x;
  int v = 1000 + l;
  // permlane32_swap(old, src): exchanges rows 0-31 <-> 32-63 of the pair
  auto pair = __builtin_amdgcn_permlane32_swap(v, v + 100000, false, false);
  out[l * 2] = pair[0];
  out[l * 2 + 1] = pair[1];
}

int main() {
  float *A, *B, *C;
  hipMallocManaged(&A, 32 * 16 * 4);
  hipMallocManaged(&B, 16 * 32 * 4);
  hipMallocManaged(&C, 32 * 32 * 4);
  for (int i = 0; i < 32 * 16; ++i) A[i] = (i % 23) * 0.25f - 2.f;
  for (int i = 0; i < 16 * 32; ++i) B[i] = ((i * 7) % 19) * 0.5f - 4.f;
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, A, B, C);
  hipDeviceSynchronize();
  int bad = 0;
  for (int m = 0; m < 32 && bad < 6; ++m)
    for (int n = 0; n < 32 && bad < 6; ++n) {
      float want = 0;
      for (int k = 0; k < 16; ++k) want += A[m * 16 + k] * B[k * 32 + n];
      if (fabsf(want - C[m * 32 + n]) > 0.5f) {
        printf("C[%d][%d]=%f want %f\n", m, n, C[m * 32 + n], want);
        ++bad;
      }
    }
  printf(bad ? "MFMA32 LAYOUT WRONG (%d+ mismatches)\n" : "MFMA32 layout OK\n",
         bad);

  int* P;
  hipMallocManaged(&P, 64 * 2 * 4);
  hipLaunchKernelGGL(probe_permlane, dim3(1), dim3(64), 0, 0, P);
  hipDeviceSynchronize();
  printf("permlane32_swap lane0=[%d,%d] lane32=[%d,%d] lane1=[%d,%d]\n",
         P[0], P[1], P[64], P[65], P[2], P[3]);
  return 0;
}
