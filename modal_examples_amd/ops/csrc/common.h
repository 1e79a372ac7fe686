// Common device helpers for the gfx950 (CDNA4 / MI355X) kernel suite.
//
// Design constants per /opt/skills/guides/cdna_hip_programming.md:
//  - wavefront = 64 lanes (never 32)
//  - MFMA bf16: __builtin_amdgcn_mfma_f32_16x16x32_bf16, frag = 8 bf16 (4 VGPR)
//    A: lane holds A[l&15][(l>>4)*8 + j]   (row, contiguous k-chunk)
//    B: lane holds B[(l>>4)*8 + j][l&15]   (contiguous k-chunk, col)
//    C/D: lane holds C[(l>>4)*4 + r][l&15] (r = reg index 0..3)
//  - LDS 32 banks x 4B; 160 KiB/CU; pad rows +8 bf16 to break power-of-2 strides
//  - vectorize bf16 global access as short4/short8 (8-16 B/lane)
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

typedef __attribute__((ext_vector_type(8))) short bf16x8;   // MFMA A/B fragment
typedef __attribute__((ext_vector_type(4))) float f32x4;    // MFMA C/D fragment
typedef __attribute__((ext_vector_type(4))) short bf16x4;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) int i32x4;

DEV_INLINE float bf2f(short u) {
  union { float f; unsigned int i; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}

DEV_INLINE short f2bf(float f) {
  union { float f; unsigned int i; } c;
  c.f = f;
  // round-to-nearest-even
  unsigned int rounding_bias = 0x7FFF + ((c.i >> 16) & 1);
  return (short)((c.i + rounding_bias) >> 16);
}

// Wave-group reductions: reduce over the 16 lanes of an MFMA column group
// (xor strides 1,2,4,8 stay inside a 16-lane group).
DEV_INLINE float group16_max(float v) {
#pragma unroll
  for (int s = 1; s < 16; s <<= 1) v = fmaxf(v, __shfl_xor(v, s, WAVE));
  return v;
}

DEV_INLINE float group16_sum(float v) {
#pragma unroll
  for (int s = 1; s < 16; s <<= 1) v += __shfl_xor(v, s, WAVE);
  return v;
}

DEV_INLINE float wave_max(float v) {
#pragma unroll
  for (int s = 1; s < 64; s <<= 1) v = fmaxf(v, __shfl_xor(v, s, WAVE));
  return v;
}

DEV_INLINE float wave_sum(float v) {
#pragma unroll
  for (int s = 1; s < 64; s <<= 1) v += __shfl_xor(v, s, WAVE);
  return v;
}

// Grid sizing for memory-bound elementwise/reduction kernels:
// cap at ~2048 blocks and grid-stride (guide §6 Guideline 11).
static inline int elementwise_grid(long long total, int block, int per_thread = 8) {
  long long blocks = (total + (long long)block * per_thread - 1) / ((long long)block * per_thread);
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(_e),        \
              __FILE__, __LINE__);                                             \
      abort();                                                                 \
    }                                                                          \
  } while (0)
