// Paged-KV decode attention (single query token per sequence), bf16, gfx950.
//
// Serves K6: the vLLM/SGLang-style PagedAttention that the
// reference's serving examples lean on (llm-serving/vllm_inference.py:158-209).
//
// MI355X-first design: decode is HBM-bound (reads the whole KV cache once per
// step), so the kernel is organized around 16-byte-per-lane vector loads and
// GQA amortization — each workgroup loads the KV rows of ONE kv-head and scores
// all G = Hq/Hkv query heads of its group against them, so KV bytes are read
// once per group instead of once per query head.
//
//   grid = (Hkv, B, SPLITS), block = 256 (4 waves); SPLITS > 1 engages
//   flash-decoding: each split covers a kv range and writes an (m, l, acc)
//   partial to a workspace, merged by a second kernel — small-batch decode
//   otherwise uses only B*Hkv workgroups of a 256-CU chip
//   lane split: D/8 lanes cover one KV row (16B each) → ROWS=64/(D/8) rows
//   per wave step; each row-group keeps an online-softmax partial (m, l, acc)
//   per query head; partials merge through LDS at the end (flash-decoding
//   style merge with max realignment).
//
// KV cache layout: [num_blocks, Hkv, block_size, D] (block_table int32
// [B, max_blocks]); pass block_table = nullptr for a contiguous [B, Hkv, S, D]
// cache (then block_size must be >= max seq len).
#include "common.h"

#include <cstdio>
#include <type_traits>

#define DEC_THREADS 256
#define DEC_WAVES 4
#define MAX_G 8

typedef __attribute__((ext_vector_type(2))) unsigned int u32x2d;

// OCP e4m3 -> f32 via the gfx950 HW converter (4 fp8 per dword, constant
// lane select required by the builtin).
DEV_INLINE void fp8x4_to_f32(unsigned w, float* out4) {
  out4[0] = __builtin_amdgcn_cvt_f32_fp8(w, 0);
  out4[1] = __builtin_amdgcn_cvt_f32_fp8(w, 1);
  out4[2] = __builtin_amdgcn_cvt_f32_fp8(w, 2);
  out4[3] = __builtin_amdgcn_cvt_f32_fp8(w, 3);
}

// GT = compile-time GQA group size: register arrays (qf/accv) and the LDS
// merge buffers are sized by GT, not MAX_G — at Llama's G=4 this halves the
// VGPR/LDS footprint (205 VGPR/66 KB -> more waves + blocks per CU; decode
// is HBM-latency-bound so occupancy IS the bandwidth lever, probe_vmcnt_r2).
// FP8: KV cache stored as OCP e4m3 (the vllm_low_latency FP8 role) — halves
// KV bytes per token; each lane loads 8 fp8 (8 B) instead of 8 bf16 (16 B)
// and converts through the HW fp8 pipe.
template <int D, int GT, bool FP8>
__global__ __launch_bounds__(DEC_THREADS) void paged_decode_kernel(
    const short* __restrict__ Q,       // [B, Hq, D]
    const short* __restrict__ Kc,      // cache, layout above
    const short* __restrict__ Vc,
    const int* __restrict__ block_table,  // [B, max_blocks] or nullptr
    const int* __restrict__ seq_lens,     // [B]
    short* __restrict__ O,             // [B, Hq, D]
    float* __restrict__ WS,            // [B, Hq, splits, D+2] or nullptr
    int B, int Hq, int Hkv, int block_size, int max_blocks, int splits,
    float scale) {
  constexpr int LPR = D / 8;          // lanes per KV row (16B chunks)
  constexpr int ROWS = WAVE / LPR;    // KV rows per wave step
  constexpr int NPART = DEC_WAVES * ROWS;  // softmax partials to merge

  const int hkv = blockIdx.x;
  const int b = blockIdx.y;
  const int split = blockIdx.z;
  const int G = Hq / Hkv;
  const int S_total = seq_lens[b];
  // per-sequence balanced split range
  const int chunk = (S_total + splits - 1) / splits;
  const int kv_lo = split * chunk;
  const int S = min(S_total, kv_lo + chunk);

  const int tid = threadIdx.x;
  const int w = tid / WAVE;
  const int l = tid % WAVE;
  const int slot = l % LPR;           // which 16B chunk of the row
  const int rg = l / LPR;             // row-group within the wave
  const int part = w * ROWS + rg;     // global partial index

  // Q fragments: per head g, this lane's 8 bf16 of q (its slot), as f32
  float qf[GT][8];
#pragma unroll
  for (int g = 0; g < GT; ++g) {
    if (g < G) {
      const short* qp = &Q[(((long long)b * Hq) + hkv * G + g) * D + slot * 8];
      bf16x8 q8 = *(const bf16x8*)qp;
#pragma unroll
      for (int j = 0; j < 8; ++j) qf[g][j] = bf2f(q8[j]);
    }
  }

  float m_run[GT], l_run[GT], accv[GT][8];
#pragma unroll
  for (int g = 0; g < GT; ++g) {
    m_run[g] = -1e30f;
    l_run[g] = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) accv[g][j] = 0.f;
  }

  // ---- sweep this split's KV rows of (b, hkv) ----
  // software-pipelined: the NEXT row's K/V loads issue before computing the
  // current row, hiding the ~300-cycle HBM latency under the dot products
  auto row_addr = [&](int kv) -> long long {
    if (block_table != nullptr) {
      int blk = block_table[(long long)b * max_blocks + kv / block_size];
      return (((long long)blk * Hkv + hkv) * block_size + kv % block_size) * D;
    }
    return (((long long)b * Hkv + hkv) * (long long)block_size + kv) * D;
  };
  // depth-2 pipeline with NAMED registers (a runtime-indexed ring spills to
  // scratch — measured 3.5x slower; guide rule #20).  kv_t is 16 B of bf16
  // or 8 B of fp8 — 8 elements per lane either way.
  typedef typename std::conditional<FP8, u32x2d, bf16x8>::type kv_t;
  const unsigned char* Kc8 = (const unsigned char*)Kc;
  const unsigned char* Vc8 = (const unsigned char*)Vc;
  auto load_kv = [&](const short* base16, const unsigned char* base8,
                     long long elem_off) -> kv_t {
    if (FP8) {
      return *(const kv_t*)&base8[elem_off];
    }
    return *(const kv_t*)&base16[elem_off];
  };
  auto to_f32 = [&](kv_t r, float* out8) {
    if (FP8) {
      u32x2d w = *(u32x2d*)&r;
      fp8x4_to_f32(w[0], out8);
      fp8x4_to_f32(w[1], out8 + 4);
    } else {
      bf16x8 b = *(bf16x8*)&r;
#pragma unroll
      for (int j = 0; j < 8; ++j) out8[j] = bf2f(b[j]);
    }
  };
  kv_t kA = {}, vA = {}, kB = {}, vB = {};
  int kv0 = kv_lo + part;
  if (kv0 < S) {
    long long o0 = row_addr(kv0) + slot * 8;
    kA = load_kv(Kc, Kc8, o0);
    vA = load_kv(Vc, Vc8, o0);
  }
  if (kv0 + NPART < S) {
    long long o1 = row_addr(kv0 + NPART) + slot * 8;
    kB = load_kv(Kc, Kc8, o1);
    vB = load_kv(Vc, Vc8, o1);
  }
  for (int kv = kv0; kv < S; kv += NPART) {
    kv_t k8 = kA, v8 = vA;
    kA = kB;
    vA = vB;
    if (kv + 2 * NPART < S) {
      long long offn = row_addr(kv + 2 * NPART) + slot * 8;
      kB = load_kv(Kc, Kc8, offn);
      vB = load_kv(Vc, Vc8, offn);
    }
    float kfl[8], vfl[8];
    to_f32(k8, kfl);
    to_f32(v8, vfl);

#pragma unroll
    for (int g = 0; g < GT; ++g) {
      if (g >= G) break;
      float d = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) d += qf[g][j] * kfl[j];
      // reduce across the LPR lanes of this row
#pragma unroll
      for (int s_ = 1; s_ < LPR; s_ <<= 1) d += __shfl_xor(d, s_, WAVE);
      d *= scale;
      float m_new = fmaxf(m_run[g], d);
      float rs = __expf(m_run[g] - m_new);
      float p = __expf(d - m_new);
      l_run[g] = l_run[g] * rs + p;
      m_run[g] = m_new;
#pragma unroll
      for (int j = 0; j < 8; ++j) accv[g][j] = accv[g][j] * rs + p * vfl[j];
    }
  }

  // ---- merge partials through LDS ----
  __shared__ float sm[GT][NPART];
  __shared__ float sl[GT][NPART];
  __shared__ float sacc[GT][NPART][D];

#pragma unroll
  for (int g = 0; g < GT; ++g) {
    if (g >= G) break;
    if (slot == 0) {
      sm[g][part] = m_run[g];
      sl[g][part] = l_run[g];
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) sacc[g][part][slot * 8 + j] = accv[g][j];
  }
  __syncthreads();

  // threads cover (g, d) output elements
  for (int gd = tid; gd < G * D; gd += DEC_THREADS) {
    int g = gd / D, d = gd % D;
    float m_star = -1e30f;
    for (int p_ = 0; p_ < NPART; ++p_) m_star = fmaxf(m_star, sm[g][p_]);
    float num = 0.f, den = 0.f;
    for (int p_ = 0; p_ < NPART; ++p_) {
      float f = __expf(sm[g][p_] - m_star);
      num += f * sacc[g][p_][d];
      den += f * sl[g][p_];
    }
    int h = hkv * G + g;
    if (WS == nullptr) {
      O[(((long long)b * Hq) + h) * D + d] =
          f2bf(den > 0.f ? num / den : 0.f);
    } else {
      float* w = WS + ((((long long)b * Hq) + h) * splits + split) * (D + 2);
      w[d] = num;  // unnormalized, at this split's m_star
      if (d == 0) {
        w[D] = m_star;
        w[D + 1] = den;
      }
    }
  }
}

// merge the split partials: O[b,h,:] = sum_s e^{m_s - m*} num_s / sum den_s
__global__ __launch_bounds__(256) void decode_merge_kernel(
    const float* __restrict__ WS, short* __restrict__ O, int B, int Hq, int D,
    int splits) {
  int bh = blockIdx.x;
  const float* base = WS + (long long)bh * splits * (D + 2);
  __shared__ float m_star_s;
  if (threadIdx.x == 0) {
    float m = -1e30f;
    for (int s_ = 0; s_ < splits; ++s_) m = fmaxf(m, base[s_ * (D + 2) + D]);
    m_star_s = m;
  }
  __syncthreads();
  float m_star = m_star_s;
  for (int d = threadIdx.x; d < D; d += 256) {
    float num = 0.f, den_acc = 0.f;
    for (int s_ = 0; s_ < splits; ++s_) {
      const float* w = base + s_ * (D + 2);
      float f = __expf(w[D] - m_star);
      num += f * w[d];
      den_acc += f * w[D + 1];
    }
    O[(long long)bh * D + d] = f2bf(den_acc > 0.f ? num / den_acc : 0.f);
  }
}

extern "C" void paged_decode_bf16(const void* q, const void* kc, const void* vc,
                                  const int* block_table, const int* seq_lens,
                                  void* o, float* ws, int B, int Hq, int Hkv,
                                  int D, int block_size, int max_blocks,
                                  int splits, float scale, int kv_fp8,
                                  hipStream_t stream) {
  if (Hq / Hkv > MAX_G) {
    fprintf(stderr, "paged_decode_bf16: GQA group %d > %d\n", Hq / Hkv, MAX_G);
    abort();
  }
  if (splits < 1) splits = 1;
  dim3 grid(Hkv, B, splits);
  dim3 block(DEC_THREADS);
  float* ws_arg = splits > 1 ? ws : nullptr;
  const int G = Hq / Hkv;
#define DLAUNCH(DD, GG, F8)                                                  \
  hipLaunchKernelGGL((paged_decode_kernel<DD, GG, F8>), grid, block, 0,      \
                     stream, (const short*)q, (const short*)kc,              \
                     (const short*)vc, block_table, seq_lens, (short*)o,     \
                     ws_arg, B, Hq, Hkv, block_size, max_blocks, splits,     \
                     scale)
#define DGROUP(DD, F8)                                                      \
  do {                                                                       \
    if (G == 1) DLAUNCH(DD, 1, F8);                                          \
    else if (G == 2) DLAUNCH(DD, 2, F8);                                     \
    else if (G <= 4) DLAUNCH(DD, 4, F8);                                     \
    else DLAUNCH(DD, 8, F8);                                                 \
  } while (0)
#define DDISPATCH(DD)                                                        \
  do {                                                                       \
    if (kv_fp8) DGROUP(DD, true); else DGROUP(DD, false);                    \
  } while (0)
  if (D == 64) {
    DDISPATCH(64);
  } else if (D == 128) {
    DDISPATCH(128);
  } else {
    fprintf(stderr, "paged_decode_bf16: unsupported head_dim %d\n", D);
    abort();
  }
#undef DDISPATCH
#undef DGROUP
#undef DLAUNCH
  if (splits > 1) {
    hipLaunchKernelGGL(decode_merge_kernel, dim3(B * Hq), dim3(256), 0,
                       stream, ws, (short*)o, B, Hq, D, splits);
  }
}
