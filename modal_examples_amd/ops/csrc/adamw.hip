// Fused AdamW, gfx950.  Serves K9: the optimizer step of the
// Dreambooth-LoRA fine-tune (reference trigger:
// 06_gpu_and_ml/dreambooth/diffusers_lora_finetune.py:309-339, accelerate's
// fused Adam).  One pass over {param, grad, m, v} per step; params/grads may be
// bf16 (with f32 moments) or f32.  Bias correction folded into the step size
// on the host side would lose per-step exactness vs torch — computed here.
#include "common.h"

#define AW_BLOCK 256

template <typename PT>  // short (bf16) or float
__global__ __launch_bounds__(AW_BLOCK) void adamw_kernel(
    PT* __restrict__ P, const PT* __restrict__ Gr, float* __restrict__ M,
    float* __restrict__ V, long long n, float lr, float beta1, float beta2,
    float eps, float wd, float bc1, float bc2) {
  long long stride = (long long)gridDim.x * AW_BLOCK;
  for (long long i = (long long)blockIdx.x * AW_BLOCK + threadIdx.x; i < n;
       i += stride) {
    float p, g;
    if constexpr (sizeof(PT) == 2) {
      p = bf2f(P[i]);
      g = bf2f(Gr[i]);
    } else {
      p = ((const float*)P)[i];
      g = ((const float*)Gr)[i];
    }
    float m = M[i] = beta1 * M[i] + (1.f - beta1) * g;
    float v = V[i] = beta2 * V[i] + (1.f - beta2) * g * g;
    float mhat = m / bc1;
    float vhat = v / bc2;
    p -= lr * (mhat / (sqrtf(vhat) + eps) + wd * p);
    if constexpr (sizeof(PT) == 2)
      P[i] = f2bf(p);
    else
      ((float*)P)[i] = p;
  }
}

extern "C" void adamw_step(void* p, const void* g, float* m, float* v,
                           long long n, float lr, float beta1, float beta2,
                           float eps, float wd, int step, int is_bf16,
                           hipStream_t stream) {
  float bc1 = 1.f - powf(beta1, (float)step);
  float bc2 = 1.f - powf(beta2, (float)step);
  int grid = elementwise_grid(n, AW_BLOCK, 4);
  if (is_bf16) {
    hipLaunchKernelGGL((adamw_kernel<short>), dim3(grid), dim3(AW_BLOCK), 0,
                       stream, (short*)p, (const short*)g, m, v, n, lr, beta1,
                       beta2, eps, wd, bc1, bc2);
  } else {
    hipLaunchKernelGGL((adamw_kernel<float>), dim3(grid), dim3(AW_BLOCK), 0,
                       stream, (float*)p, (const float*)g, m, v, n, lr, beta1,
                       beta2, eps, wd, bc1, bc2);
  }
}
