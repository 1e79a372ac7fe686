// Standalone ds_read_b64_tr_b16 semantics probe (compile+run on the GPU box):
//   hipcc --offload-arch=gfx950 -O2 scripts/probe_tr16.cpp -o /tmp/probe && /tmp/probe
// LDS holds lds[i] = i; three address patterns reveal the lane/data mapping.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(4))) short s4;
typedef __attribute__((address_space(3))) s4* lds_p;

__global__ void probe(short* out, const int* addrs) {
  __shared__ short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  int l = threadIdx.x;
  if (l < 64) {
    s4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds_p)&lds[addrs[l]]);
    for (int j = 0; j < 4; ++j) out[l * 4 + j] = v[j];
  }
}

int run(const char* name, int* addrs_h) {
  int* addrs_d;
  short* out_d;
  hipMalloc(&addrs_d, 64 * sizeof(int));
  hipMalloc(&out_d, 256 * sizeof(short));
  hipMemcpy(addrs_d, addrs_h, 64 * sizeof(int), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, out_d, addrs_d);
  short out_h[256];
  hipMemcpy(out_h, out_d, sizeof(out_h), hipMemcpyDeviceToHost);
  printf("== %s\n", name);
  for (int l = 0; l < 64; l += 1) {
    if (l % 16 < 3 || l % 16 == 15) {
      printf("lane %2d addr %4d -> [%4d %4d %4d %4d]\n", l, addrs_h[l],
             out_h[l * 4], out_h[l * 4 + 1], out_h[l * 4 + 2], out_h[l * 4 + 3]);
    }
  }
  hipFree(addrs_d);
  hipFree(out_d);
  return 0;
}

int main() {
  int a[64];
  for (int l = 0; l < 64; ++l) a[l] = l;
  run("linear addr=l", a);
  for (int l = 0; l < 64; ++l) a[l] = (l * 5) % 64;
  run("scrambled addr=(5l)%64", a);
  for (int l = 0; l < 64; ++l) a[l] = 128;
  run("uniform addr=128", a);
  for (int l = 0; l < 64; ++l) a[l] = (l % 16) * 16;  // 16 different tiles per group?
  run("addr=(l%16)*16", a);
  return 0;
}
