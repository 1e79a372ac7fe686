// Fused elementwise kernels, bf16, gfx950.  All HBM-bound → bf16x8 vector
// access, grid-stride, capped grid (guide §6 G11/G13).
//
// Serves: K4 sampler step + CFG combine (the per-step
// elementwise work inside the reference's diffusion pipelines,
// text_to_image.py:114-120), SiLU-mul / GEGLU for transformer FFNs (K2 fusion
// family), residual adds, RoPE with host-precomputed cos/sin tables
// (guide Appendix B: never compute trig on-device).
#include "common.h"

#define EW_BLOCK 256

// ---------------------------------------------------------------- CFG + Euler step
// Classifier-free guidance combine and ancestral-free Euler update in one pass:
//   eps = eps_u + g*(eps_c - eps_u);  x_next = x + (sig_next - sig) * eps
// (one kernel per denoise step instead of 4 reads + 3 writes).

__global__ __launch_bounds__(EW_BLOCK) void cfg_euler_kernel(
    const short* __restrict__ Xt, const short* __restrict__ EpsC,
    const short* __restrict__ EpsU, short* __restrict__ Xn, float guidance,
    float dsigma, long long n) {
  long long stride = (long long)gridDim.x * EW_BLOCK * 8;
  for (long long i = ((long long)blockIdx.x * EW_BLOCK + threadIdx.x) * 8;
       i + 8 <= n; i += stride) {
    bf16x8 x = *(const bf16x8*)&Xt[i];
    bf16x8 ec = *(const bf16x8*)&EpsC[i];
    bf16x8 o;
    if (EpsU != nullptr) {
      bf16x8 eu = *(const bf16x8*)&EpsU[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float e = bf2f(eu[j]) + guidance * (bf2f(ec[j]) - bf2f(eu[j]));
        o[j] = f2bf(bf2f(x[j]) + dsigma * e);
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = f2bf(bf2f(x[j]) + dsigma * bf2f(ec[j]));
    }
    *(bf16x8*)&Xn[i] = o;
  }
  // tail
  long long full = (n / 8) * 8;
  for (long long i = full + (long long)blockIdx.x * EW_BLOCK + threadIdx.x;
       i < n; i += (long long)gridDim.x * EW_BLOCK) {
    float e = bf2f(EpsC[i]);
    if (EpsU != nullptr) {
      float eu = bf2f(EpsU[i]);
      e = eu + guidance * (e - eu);
    }
    Xn[i] = f2bf(bf2f(Xt[i]) + dsigma * e);
  }
}

extern "C" void cfg_euler_bf16(const void* xt, const void* eps_c,
                               const void* eps_u, void* xn, float guidance,
                               float dsigma, long long n, hipStream_t stream) {
  int grid = elementwise_grid(n, EW_BLOCK);
  hipLaunchKernelGGL(cfg_euler_kernel, dim3(grid), dim3(EW_BLOCK), 0, stream,
                     (const short*)xt, (const short*)eps_c,
                     (const short*)eps_u, (short*)xn, guidance, dsigma, n);
}

// ---------------------------------------------------------------- SiLU-mul (SwiGLU)
// y = silu(a) * b  — Llama FFN gate; a,b are the two halves of the gate_up proj.

__global__ __launch_bounds__(EW_BLOCK) void silu_mul_kernel(
    const short* __restrict__ A, const short* __restrict__ Bv,
    short* __restrict__ Y, long long n) {
  long long stride = (long long)gridDim.x * EW_BLOCK * 8;
  for (long long i = ((long long)blockIdx.x * EW_BLOCK + threadIdx.x) * 8;
       i + 8 <= n; i += stride) {
    bf16x8 a = *(const bf16x8*)&A[i];
    bf16x8 b = *(const bf16x8*)&Bv[i];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(a[j]);
      o[j] = f2bf(f / (1.f + __expf(-f)) * bf2f(b[j]));
    }
    *(bf16x8*)&Y[i] = o;
  }
  long long full = (n / 8) * 8;
  for (long long i = full + (long long)blockIdx.x * EW_BLOCK + threadIdx.x;
       i < n; i += (long long)gridDim.x * EW_BLOCK) {
    float f = bf2f(A[i]);
    Y[i] = f2bf(f / (1.f + __expf(-f)) * bf2f(Bv[i]));
  }
}

extern "C" void silu_mul_bf16(const void* a, const void* b, void* y,
                              long long n, hipStream_t stream) {
  int grid = elementwise_grid(n, EW_BLOCK);
  hipLaunchKernelGGL(silu_mul_kernel, dim3(grid), dim3(EW_BLOCK), 0, stream,
                     (const short*)a, (const short*)b, (short*)y, n);
}

// ---------------------------------------------------------------- GEGLU
// y = gelu(a) * b — SDXL transformer FFN (diffusers GEGLU). tanh approximation.

__global__ __launch_bounds__(EW_BLOCK) void geglu_kernel(
    const short* __restrict__ A, const short* __restrict__ Bv,
    short* __restrict__ Y, long long n) {
  long long stride = (long long)gridDim.x * EW_BLOCK * 8;
  const float k0 = 0.7978845608028654f, k1 = 0.044715f;
  for (long long i = ((long long)blockIdx.x * EW_BLOCK + threadIdx.x) * 8;
       i + 8 <= n; i += stride) {
    bf16x8 a = *(const bf16x8*)&A[i];
    bf16x8 b = *(const bf16x8*)&Bv[i];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(a[j]);
      float g = 0.5f * f * (1.f + tanhf(k0 * (f + k1 * f * f * f)));
      o[j] = f2bf(g * bf2f(b[j]));
    }
    *(bf16x8*)&Y[i] = o;
  }
  long long full = (n / 8) * 8;
  for (long long i = full + (long long)blockIdx.x * EW_BLOCK + threadIdx.x;
       i < n; i += (long long)gridDim.x * EW_BLOCK) {
    float f = bf2f(A[i]);
    float g = 0.5f * f * (1.f + tanhf(k0 * (f + k1 * f * f * f)));
    Y[i] = f2bf(g * bf2f(Bv[i]));
  }
}

extern "C" void geglu_bf16(const void* a, const void* b, void* y, long long n,
                           hipStream_t stream) {
  int grid = elementwise_grid(n, EW_BLOCK);
  hipLaunchKernelGGL(geglu_kernel, dim3(grid), dim3(EW_BLOCK), 0, stream,
                     (const short*)a, (const short*)b, (short*)y, n);
}

// ---------------------------------------------------------------- residual add

__global__ __launch_bounds__(EW_BLOCK) void add_kernel(
    const short* __restrict__ A, const short* __restrict__ Bv,
    short* __restrict__ Y, long long n) {
  long long stride = (long long)gridDim.x * EW_BLOCK * 8;
  for (long long i = ((long long)blockIdx.x * EW_BLOCK + threadIdx.x) * 8;
       i + 8 <= n; i += stride) {
    bf16x8 a = *(const bf16x8*)&A[i];
    bf16x8 b = *(const bf16x8*)&Bv[i];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(a[j]) + bf2f(b[j]));
    *(bf16x8*)&Y[i] = o;
  }
  long long full = (n / 8) * 8;
  for (long long i = full + (long long)blockIdx.x * EW_BLOCK + threadIdx.x;
       i < n; i += (long long)gridDim.x * EW_BLOCK)
    Y[i] = f2bf(bf2f(A[i]) + bf2f(Bv[i]));
}

extern "C" void add_bf16(const void* a, const void* b, void* y, long long n,
                         hipStream_t stream) {
  int grid = elementwise_grid(n, EW_BLOCK);
  hipLaunchKernelGGL(add_kernel, dim3(grid), dim3(EW_BLOCK), 0, stream,
                     (const short*)a, (const short*)b, (short*)y, n);
}

// ---------------------------------------------------------------- RoPE
// q/k: [B, H, S, D] bf16; cos/sin: [S, D/2] f32 host-precomputed (G13/App.B:
// on-device trig turns this memory-bound op VALU-bound). Llama rotate-half
// convention: (x1, x2) = (x[..D/2], x[D/2..]); x1' = x1*cos - x2*sin, etc.
// In-place over q and k in one launch.

__global__ __launch_bounds__(EW_BLOCK) void rope_kernel(
    short* __restrict__ Qk, const float* __restrict__ Cos,
    const float* __restrict__ Sin, long long B, int H, int S, int D,
    long long bst, long long hst, long long sst,
    const int* __restrict__ positions /*nullable [B*S], per-seq offsets*/) {
  // one wave handles one (b, h, s) row; lanes cover D/2 rotation pairs
  long long rows = B * H * S;
  int l = threadIdx.x % WAVE;
  for (long long row = blockIdx.x * (EW_BLOCK / WAVE) + threadIdx.x / WAVE;
       row < rows; row += (long long)gridDim.x * (EW_BLOCK / WAVE)) {
    int s = (int)(row % S);
    int h = (int)((row / S) % H);
    long long b = row / ((long long)H * S);
    int pos = positions != nullptr ? positions[b * S + s] : s;
    short* x = Qk + b * bst + h * hst + s * sst;
    for (int i = l * 2; i < D / 2; i += WAVE * 2) {
      float c0 = Cos[(long long)pos * (D / 2) + i];
      float s0 = Sin[(long long)pos * (D / 2) + i];
      float c1 = Cos[(long long)pos * (D / 2) + i + 1];
      float s1 = Sin[(long long)pos * (D / 2) + i + 1];
      float x0 = bf2f(x[i]), x2 = bf2f(x[i + D / 2]);
      float x1 = bf2f(x[i + 1]), x3 = bf2f(x[i + 1 + D / 2]);
      x[i] = f2bf(x0 * c0 - x2 * s0);
      x[i + D / 2] = f2bf(x2 * c0 + x0 * s0);
      x[i + 1] = f2bf(x1 * c1 - x3 * s1);
      x[i + 1 + D / 2] = f2bf(x3 * c1 + x1 * s1);
    }
  }
}

extern "C" void rope_bf16(void* qk, const float* cosv, const float* sinv,
                          long long B, int H, int S, int D, long long bst,
                          long long hst, long long sst, const int* positions,
                          hipStream_t stream) {
  long long rows = B * (long long)H * S;
  int grid = (int)((rows + 3) / 4);
  if (grid > 4096) grid = 4096;
  hipLaunchKernelGGL(rope_kernel, dim3(grid), dim3(EW_BLOCK), 0, stream,
                     (short*)qk, cosv, sinv, B, H, S, D, bst, hst, sst,
                     positions);
}

// ---------------------------------------------------------------- graph-capturable
// Device-scalar variants for hipGraph capture of the denoise loop: the sigma
// schedule lives in device memory and a device step counter advances INSIDE
// the captured graph, so one capture serves every step (SURVEY.md §7 phase 3:
// "hipGraph capture of the denoise loop").

__global__ __launch_bounds__(EW_BLOCK) void scale_in_dev_kernel(
    const short* __restrict__ X, short* __restrict__ Y,
    const float* __restrict__ sigmas, const long long* __restrict__ step,
    long long n) {
  float s = sigmas[(int)*step];
  float c = rsqrtf(s * s + 1.f);
  long long stride = (long long)gridDim.x * EW_BLOCK * 8;
  for (long long i = ((long long)blockIdx.x * EW_BLOCK + threadIdx.x) * 8;
       i + 8 <= n; i += stride) {
    bf16x8 x = *(const bf16x8*)&X[i];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(x[j]) * c);
    *(bf16x8*)&Y[i] = o;
  }
  long long full = (n / 8) * 8;
  for (long long i = full + (long long)blockIdx.x * EW_BLOCK + threadIdx.x;
       i < n; i += (long long)gridDim.x * EW_BLOCK)
    Y[i] = f2bf(bf2f(X[i]) * c);
}

__global__ __launch_bounds__(EW_BLOCK) void cfg_euler_dev_kernel(
    const short* __restrict__ Xt, const short* __restrict__ EpsC,
    const short* __restrict__ EpsU, short* __restrict__ Xn,
    const float* __restrict__ sigmas, const long long* __restrict__ step,
    float guidance, long long n) {
  int st = (int)*step;
  float dsigma = sigmas[st + 1] - sigmas[st];
  long long stride = (long long)gridDim.x * EW_BLOCK * 8;
  for (long long i = ((long long)blockIdx.x * EW_BLOCK + threadIdx.x) * 8;
       i + 8 <= n; i += stride) {
    bf16x8 x = *(const bf16x8*)&Xt[i];
    bf16x8 ec = *(const bf16x8*)&EpsC[i];
    bf16x8 o;
    if (EpsU != nullptr) {
      bf16x8 eu = *(const bf16x8*)&EpsU[i];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float e = bf2f(eu[j]) + guidance * (bf2f(ec[j]) - bf2f(eu[j]));
        o[j] = f2bf(bf2f(x[j]) + dsigma * e);
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = f2bf(bf2f(x[j]) + dsigma * bf2f(ec[j]));
    }
    *(bf16x8*)&Xn[i] = o;
  }
  long long full = (n / 8) * 8;
  for (long long i = full + (long long)blockIdx.x * EW_BLOCK + threadIdx.x;
       i < n; i += (long long)gridDim.x * EW_BLOCK) {
    float e = bf2f(EpsC[i]);
    if (EpsU != nullptr) {
      float eu = bf2f(EpsU[i]);
      e = eu + guidance * (e - eu);
    }
    Xn[i] = f2bf(bf2f(Xt[i]) + dsigma * e);
  }
}

__global__ void advance_step_kernel(long long* step) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *step += 1;
}

extern "C" void scale_in_dev_bf16(const void* x, void* y, const float* sigmas,
                                  const long long* step, long long n,
                                  hipStream_t stream) {
  int grid = elementwise_grid(n, EW_BLOCK);
  hipLaunchKernelGGL(scale_in_dev_kernel, dim3(grid), dim3(EW_BLOCK), 0,
                     stream, (const short*)x, (short*)y, sigmas, step, n);
}

extern "C" void cfg_euler_dev_bf16(const void* xt, const void* eps_c,
                                   const void* eps_u, void* xn,
                                   const float* sigmas, const long long* step,
                                   float guidance, long long n,
                                   hipStream_t stream) {
  int grid = elementwise_grid(n, EW_BLOCK);
  hipLaunchKernelGGL(cfg_euler_dev_kernel, dim3(grid), dim3(EW_BLOCK), 0,
                     stream, (const short*)xt, (const short*)eps_c,
                     (const short*)eps_u, (short*)xn, sigmas, step, guidance,
                     n);
}

extern "C" void advance_step(long long* step, hipStream_t stream) {
  hipLaunchKernelGGL(advance_step_kernel, dim3(1), dim3(64), 0, stream, step);
}

// ---------------------------------------------------------------- fused-
// projection activations: the model computes ONE GEMM producing [rows, 2*I]
// (gate_up / GEGLU's ab); these kernels read the two halves IN PLACE, so the
// two .contiguous() slice copies per FFN disappear.
//   y[row, c] = act(src[row*2I + c]) * src[row*2I + I + c]

template <bool GELU>
__global__ __launch_bounds__(EW_BLOCK) void glu_fused_kernel(
    const short* __restrict__ Src, short* __restrict__ Y, long long rows,
    long long inner) {
  const float k0 = 0.7978845608028654f, k1 = 0.044715f;
  long long n = rows * inner;
  long long stride = (long long)gridDim.x * EW_BLOCK * 8;
  for (long long i = ((long long)blockIdx.x * EW_BLOCK + threadIdx.x) * 8;
       i + 8 <= n; i += stride) {
    long long row = i / inner, c = i - row * inner;  // 8-aligned within a row
    const short* base = Src + row * 2 * inner + c;
    bf16x8 a = *(const bf16x8*)base;
    bf16x8 b = *(const bf16x8*)(base + inner);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(a[j]);
      float act;
      if (GELU) {
        float t = tanhf(k0 * (f + k1 * f * f * f));
        act = 0.5f * f * (1.f + t);
      } else {
        act = f / (1.f + __expf(-f));
      }
      o[j] = f2bf(act * bf2f(b[j]));
    }
    *(bf16x8*)&Y[i] = o;
  }
  long long full = (n / 8) * 8;
  for (long long i = full + (long long)blockIdx.x * EW_BLOCK + threadIdx.x;
       i < n; i += (long long)gridDim.x * EW_BLOCK) {
    long long row = i / inner, c = i - row * inner;
    float f = bf2f(Src[row * 2 * inner + c]);
    float bb = bf2f(Src[row * 2 * inner + inner + c]);
    float act;
    if (GELU) {
      float t = tanhf(k0 * (f + k1 * f * f * f));
      act = 0.5f * f * (1.f + t);
    } else {
      act = f / (1.f + __expf(-f));
    }
    Y[i] = f2bf(act * bb);
  }
}

extern "C" void glu_fused_bf16(const void* src, void* y, long long rows,
                               long long inner, int gelu,
                               hipStream_t stream) {
  int grid = elementwise_grid(rows * inner, EW_BLOCK);
  if (gelu)
    hipLaunchKernelGGL((glu_fused_kernel<true>), dim3(grid), dim3(EW_BLOCK),
                       0, stream, (const short*)src, (short*)y, rows, inner);
  else
    hipLaunchKernelGGL((glu_fused_kernel<false>), dim3(grid), dim3(EW_BLOCK),
                       0, stream, (const short*)src, (short*)y, rows, inner);
}
