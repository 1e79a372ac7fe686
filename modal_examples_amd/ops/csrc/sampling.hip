// Fused LLM sampling, gfx950.  Serves K8: logits → token
// without materializing softmax (reference trigger: the generate loops of
// transformers/vLLM at batched_whisper.py:133, vllm_inference.py:158-209, and
// the explicit sampling loop at hp_sweep_gpt/src/model.py:148-157).
//
// Gumbel-max trick: argmax(logits/T + G_i), G_i = -log(-log(u_i)) samples the
// softmax(logits/T) distribution in ONE read of the logits row — no max pass,
// no sum pass, no CDF scan over a 128k vocab.  u_i comes from a counter-based
// hash (philox-lite) of (seed, row, i) → deterministic per seed. T=0 → argmax.
#include "common.h"

#define SMP_BLOCK 256

DEV_INLINE unsigned int hash3(unsigned int a, unsigned int b, unsigned int c) {
  // xxhash-style avalanche mix of three words
  unsigned int h = a * 0x9E3779B1u ^ b * 0x85EBCA77u ^ c * 0xC2B2AE3Du;
  h ^= h >> 15;
  h *= 0x2C1B3C6Du;
  h ^= h >> 12;
  h *= 0x297A2D39u;
  h ^= h >> 15;
  return h;
}

__global__ __launch_bounds__(SMP_BLOCK) void gumbel_sample_kernel(
    const float* __restrict__ Logits,  // [rows, V] f32 (final-layer output)
    unsigned long long* __restrict__ Keys,  // [rows] packed argmax workspace
    int rows, int V, int splits, float inv_temp, unsigned long long seed) {
  int row = blockIdx.x / splits;
  int split = blockIdx.x % splits;
  if (row >= rows) return;
  const float* lg = Logits + (long long)row * V;
  int chunk = (V + splits - 1) / splits;
  int lo = split * chunk, hi = min(V, lo + chunk);
  float best = -1e30f;
  int best_i = lo;
  for (int i = lo + threadIdx.x; i < hi; i += SMP_BLOCK) {
    float s = lg[i] * inv_temp;
    if (inv_temp != 0.f && seed != 0ull) {
      unsigned int u = hash3((unsigned int)seed, (unsigned int)(seed >> 32) ^ row, i);
      float uf = (u >> 8) * (1.f / 16777216.f) + 1e-10f;
      s += -__logf(-__logf(uf));
    }
    if (s > best || (s == best && i < best_i)) {
      best = s;
      best_i = i;
    }
  }
#pragma unroll
  for (int s_ = 1; s_ < WAVE; s_ <<= 1) {
    float ov = __shfl_xor(best, s_, WAVE);
    int oi = __shfl_xor(best_i, s_, WAVE);
    if (ov > best || (ov == best && oi < best_i)) {
      best = ov;
      best_i = oi;
    }
  }
  __shared__ float sv[SMP_BLOCK / WAVE];
  __shared__ int si[SMP_BLOCK / WAVE];
  int w = threadIdx.x / WAVE;
  if (threadIdx.x % WAVE == 0) {
    sv[w] = best;
    si[w] = best_i;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int j = 1; j < SMP_BLOCK / WAVE; ++j)
      if (sv[j] > best || (sv[j] == best && si[j] < best_i)) {
        best = sv[j];
        best_i = si[j];
      }
    // pack (orderable float, ~idx) so atomicMax picks max value, min index
    unsigned int ub = __float_as_uint(best);
    ub = (ub & 0x80000000u) ? ~ub : (ub | 0x80000000u);
    unsigned long long key =
        ((unsigned long long)ub << 32) | (unsigned int)(~best_i);
    atomicMax(Keys + row, key);
  }
}

__global__ void gumbel_unpack_kernel(const unsigned long long* __restrict__ Keys,
                                     int* __restrict__ Out, int rows) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < rows) Out[i] = (int)(~(unsigned int)(Keys[i] & 0xFFFFFFFFull));
}

extern "C" void gumbel_sample(const float* logits, unsigned long long* keys,
                              int* out, int rows, int V, float temperature,
                              unsigned long long seed, hipStream_t stream) {
  float inv_t = temperature > 0.f ? 1.f / temperature : 0.f;
  if (temperature <= 0.f) seed = 0ull;  // greedy
  // fill the chip: aim for >=512 blocks (256 CUs), split each row's vocab
  int splits = 1;
  while (rows * splits < 512 && splits * SMP_BLOCK * 4 < V) splits *= 2;
  hipLaunchKernelGGL(gumbel_sample_kernel, dim3(rows * splits),
                     dim3(SMP_BLOCK), 0, stream, logits, keys, rows, V, splits,
                     inv_t == 0.f ? 1.f : inv_t, seed);
  hipLaunchKernelGGL(gumbel_unpack_kernel, dim3((rows + 255) / 256), dim3(256),
                     0, stream, keys, out, rows);
}

// ---------------------------------------------------------------- row softmax
// Plain row softmax (f32 in/out) for probability outputs / tests: online
// single-pass per wave using running max+sum, then a normalize pass.

__global__ __launch_bounds__(SMP_BLOCK) void softmax_rows_kernel(
    const float* __restrict__ X, float* __restrict__ Y, int rows, int V) {
  int row = blockIdx.x;
  if (row >= rows) return;
  const float* x = X + (long long)row * V;
  float* y = Y + (long long)row * V;
  float m = -1e30f, l = 0.f;
  for (int i = threadIdx.x; i < V; i += SMP_BLOCK) {
    float v = x[i];
    float m_new = fmaxf(m, v);
    l = l * __expf(m - m_new) + __expf(v - m_new);
    m = m_new;
  }
  // block-combine (m, l)
  __shared__ float sm[SMP_BLOCK / WAVE], sl[SMP_BLOCK / WAVE];
#pragma unroll
  for (int s_ = 1; s_ < WAVE; s_ <<= 1) {
    float om = __shfl_xor(m, s_, WAVE);
    float ol = __shfl_xor(l, s_, WAVE);
    float mn = fmaxf(m, om);
    l = l * __expf(m - mn) + ol * __expf(om - mn);
    m = mn;
  }
  int w = threadIdx.x / WAVE;
  if (threadIdx.x % WAVE == 0) {
    sm[w] = m;
    sl[w] = l;
  }
  __syncthreads();
  float M = sm[0], L = sl[0];
  for (int j = 1; j < SMP_BLOCK / WAVE; ++j) {
    float mn = fmaxf(M, sm[j]);
    L = L * __expf(M - mn) + sl[j] * __expf(sm[j] - mn);
    M = mn;
  }
  float inv = 1.f / L;
  for (int i = threadIdx.x; i < V; i += SMP_BLOCK)
    y[i] = __expf(x[i] - M) * inv;
}

extern "C" void softmax_rows(const float* x, float* y, int rows, int V,
                             hipStream_t stream) {
  hipLaunchKernelGGL(softmax_rows_kernel, dim3(rows), dim3(SMP_BLOCK), 0,
                     stream, x, y, rows, V);
}
