// Standalone probe for ROUND-2 attention scheduling: does a chunk-level
// counted-vmcnt software pipeline beat drain-to-zero waiting on gfx950?
//
// Isolates the QUESTION from attention correctness: a loop streams KV-sized
// rows from HBM into registers and does MFMA-shaped FLOPs per chunk, under
// three schedules:
//   v0: load chunk -> wait all (vmcnt 0)  -> compute      (the v7 status quo)
//   v1: depth-2 named-register prefetch   (decode-kernel style)
//   v2: depth-3 named-register prefetch + s_waitcnt vmcnt(N) inline asm —
//       loads for chunk c+2/c+3 stay in flight while c computes
// Report GB/s each; if v2 >> v0 the full 8-phase rewrite is justified.
//
// Build:  hipcc --offload-arch=gfx950 -O3 scripts/probe_vmcnt_pipeline.cpp -o /tmp/probe_vm
// Run  :  /tmp/probe_vm   (on an MI355X box, e.g. first gpurun of round 2)
#include <hip/hip_runtime.h>

#include <cstdio>
#include <vector>

#define CHECK(x) do { auto e = (x); if (e) { printf("ERR %s @%d\n", hipGetErrorString(e), __LINE__); return 1; } } while (0)

using bf16x8 = __attribute__((__vector_size__(8 * sizeof(short)))) short;

constexpr int WAVE = 64;
constexpr int ROWS_PER_BLOCK = 4096;   // rows each workgroup streams
constexpr int ROW_BYTES = WAVE * 16;   // one bf16x8 per lane = 1 KiB/row

__device__ inline float bf2f(short b) {
  union { float f; unsigned u; } c;
  c.u = (unsigned)(unsigned short)b << 16;
  return c.f;
}

__device__ inline float fma_chunk(const bf16x8 &v, float acc) {
#pragma unroll
  for (int j = 0; j < 8; ++j) acc += bf2f(v[j]) * 1.0009765625f;
  return acc;
}

// ---- v0: wait-for-everything before compute (compiler drains vmcnt to 0)
__global__ void k_naive(const short *__restrict__ src, float *out, int iters) {
  const short *p = src + (blockIdx.x * (long long)ROWS_PER_BLOCK * WAVE * 8)
                 + threadIdx.x * 8;
  float acc = 0.f;
  for (int it = 0; it < iters; ++it)
    for (int r = 0; r < ROWS_PER_BLOCK; ++r) {
      bf16x8 v = *(const bf16x8 *)(p + (long long)r * WAVE * 8);
      acc = fma_chunk(v, acc);
    }
  if (acc == 12345.f) out[blockIdx.x] = acc;  // keep alive
}

// ---- v1: depth-2 named registers (attention_decode.hip pattern)
__global__ void k_depth2(const short *__restrict__ src, float *out, int iters) {
  const short *p = src + (blockIdx.x * (long long)ROWS_PER_BLOCK * WAVE * 8)
                 + threadIdx.x * 8;
  float acc = 0.f;
  for (int it = 0; it < iters; ++it) {
    bf16x8 a = *(const bf16x8 *)p;
    bf16x8 b = *(const bf16x8 *)(p + (long long)WAVE * 8);
    for (int r = 0; r < ROWS_PER_BLOCK; ++r) {
      bf16x8 cur = a;
      a = b;
      if (r + 2 < ROWS_PER_BLOCK)
        b = *(const bf16x8 *)(p + (long long)(r + 2) * WAVE * 8);
      acc = fma_chunk(cur, acc);
    }
  }
  if (acc == 12345.f) out[blockIdx.x] = acc;
}

// ---- v2: depth-3 + explicit counted waitcnt. The asm constrains the
// compiler from collapsing the pipeline: after issuing the load for c+3 we
// require only "at most 2 loads outstanding" before computing chunk c.
__global__ void k_depth3_counted(const short *__restrict__ src, float *out,
                                 int iters) {
  const short *p = src + (blockIdx.x * (long long)ROWS_PER_BLOCK * WAVE * 8)
                 + threadIdx.x * 8;
  float acc = 0.f;
  for (int it = 0; it < iters; ++it) {
    bf16x8 a = *(const bf16x8 *)p;
    bf16x8 b = *(const bf16x8 *)(p + (long long)WAVE * 8);
    bf16x8 c = *(const bf16x8 *)(p + (long long)2 * WAVE * 8);
    for (int r = 0; r < ROWS_PER_BLOCK; ++r) {
      bf16x8 cur = a;
      a = b;
      b = c;
      if (r + 3 < ROWS_PER_BLOCK)
        c = *(const bf16x8 *)(p + (long long)(r + 3) * WAVE * 8);
      // allow the two younger loads to stay in flight during compute
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
      acc = fma_chunk(cur, acc);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  if (acc == 12345.f) out[blockIdx.x] = acc;
}

template <typename K>
static float bench(K kern, const short *src, float *out, int blocks, int iters) {
  hipEvent_t t0, t1;
  hipEventCreate(&t0);
  hipEventCreate(&t1);
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(WAVE), 0, 0, src, out, iters);
  hipDeviceSynchronize();
  hipEventRecord(t0);
  hipLaunchKernelGGL(kern, dim3(blocks), dim3(WAVE), 0, 0, src, out, iters);
  hipEventRecord(t1);
  hipEventSynchronize(t1);
  float ms = 0.f;
  hipEventElapsedTime(&ms, t0, t1);
  return ms;
}

int main() {
  const int blocks = 2048, iters = 4;  // >> 256 CUs; ~8.6 GB read per launch
  const long long bytes = (long long)blocks * ROWS_PER_BLOCK * ROW_BYTES;
  short *src;
  float *out;
  CHECK(hipMalloc(&src, bytes));
  CHECK(hipMemset(src, 0x3f, bytes));
  CHECK(hipMalloc(&out, blocks * sizeof(float)));
  struct { const char *name; float ms; } r[3];
  r[0] = {"v0 naive (vmcnt drain)", bench(k_naive, src, out, blocks, iters)};
  r[1] = {"v1 depth-2 named regs ", bench(k_depth2, src, out, blocks, iters)};
  r[2] = {"v2 depth-3 + vmcnt(2) ", bench(k_depth3_counted, src, out, blocks, iters)};
  for (auto &x : r) {
    double gbs = (double)bytes * iters / (x.ms * 1e6);
    printf("%s : %7.3f ms  %8.1f GB/s\n", x.name, x.ms, gbs);
  }
  return 0;
}
