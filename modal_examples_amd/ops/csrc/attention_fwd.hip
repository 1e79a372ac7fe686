// Flash-attention forward, bf16, gfx950 (CDNA4 MFMA 16x16x32).
//
// Serves: K1 diffusion self/cross-attention (D=64, non-causal,
// Sk may differ from Sq — cross-attn over text conditioning), K5 Whisper encoder
// attention (D=64, non-causal), K7 LLM prefill (D=128, causal, GQA).
//
// Structure (one workgroup = 4 waves, each wave owns 32 query rows → 128 q/block):
//   per 32-wide KV block:
//     stage K[32][D] and V^T[D][32] into padded LDS (+8 bf16 rows breaks the
//     power-of-2 bank stride, guide §6 Guideline 4),
//     QK^T via mfma_f32_16x16x32_bf16 (Q frags live in registers),
//     online softmax (16-lane-group shuffle reductions — C-layout rows),
//     P staged through per-wave LDS as the next MFMA's A operand,
//     PV accumulates O in f32.
//
// MFMA fragment layouts used (verified mapping per cdna_hip_programming.md §3):
//   A[l&15][(l>>4)*8+j], B[(l>>4)*8+j][l&15], C[(l>>4)*4+r][l&15].
#include "common.h"

#include <cstdio>

#define QROWS_PER_WAVE 32
#define KVBLK 32
#define NWAVES 4
#define QBLK (QROWS_PER_WAVE * NWAVES)  // 128 query rows per workgroup
#define PPAD 8                          // +8 bf16 row padding

template <int D, bool CAUSAL>
__global__ __launch_bounds__(NWAVES * WAVE) void fa_fwd_kernel(
    const short* __restrict__ Q,  // [B, Hq, Sq, D] bf16
    const short* __restrict__ K,  // [B, Hkv, Sk, D]
    const short* __restrict__ V,  // [B, Hkv, Sk, D]
    short* __restrict__ O,        // [B, Hq, Sq, D]
    int B, int Hq, int Hkv, int Sq, int Sk, float scale) {
  constexpr int DCH = D / 32;   // k-chunks per MFMA row (QK^T k-dim = D)
  constexpr int DT = D / 16;    // output d-tiles
  constexpr int KPAD = D + PPAD;

  __shared__ alignas(16) short Ks[KVBLK][KPAD];
  __shared__ alignas(16) short Vts[D][KVBLK + PPAD];
  __shared__ alignas(16) short Ps[NWAVES][QROWS_PER_WAVE][KVBLK + PPAD];

  const int tid = threadIdx.x;
  const int w = tid / WAVE;       // wave id 0..3
  const int l = tid % WAVE;       // lane
  const int lr = l & 15;          // fragment row/col index
  const int lg = l >> 4;          // 16-lane group 0..3

  const int qblk = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const long long qbase = (((long long)b * Hq + h) * Sq) * D;
  const long long kbase = (((long long)b * Hkv + hkv) * Sk) * D;
  const int q0 = qblk * QBLK + w * QROWS_PER_WAVE;  // this wave's first q row
  const int causal_off = Sk - Sq;  // kv index aligned to the END of q (prefill)

  // ---- Q fragments in registers (one-time global load, bounds-checked) ----
  bf16x8 qf[2][DCH];
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
    int qr = q0 + mt * 16 + lr;
#pragma unroll
    for (int kc = 0; kc < DCH; ++kc) {
      if (qr < Sq) {
        qf[mt][kc] = *(const bf16x8*)&Q[qbase + (long long)qr * D + kc * 32 + lg * 8];
      } else {
        qf[mt][kc] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  }

  f32x4 acc[2][DT];
  float m_run[2][4], l_run[2][4];
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) acc[mt][dt] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_run[mt][r] = -1e30f;
      l_run[mt][r] = 0.f;
    }
  }

  int nkb = (Sk + KVBLK - 1) / KVBLK;
  if (CAUSAL) {
    // highest kv index any q-row in this BLOCK may attend to
    int max_kv = qblk * QBLK + QBLK - 1 + causal_off;
    int lim = (max_kv + KVBLK) / KVBLK;
    if (lim < nkb) nkb = lim;
  }

  for (int kb = 0; kb < nkb; ++kb) {
    // ---- stage K and V^T (all 256 threads, coalesced 16B chunks) ----
    __syncthreads();
    {
      constexpr int CPR = D / 8;              // 16B chunks per row
      constexpr int NCH = KVBLK * CPR;        // total chunks
      for (int ci = tid; ci < NCH; ci += NWAVES * WAVE) {
        int row = ci / CPR, c8 = ci % CPR;
        int kvp = kb * KVBLK + row;
        bf16x8 kv8 = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
        bf16x8 vv8 = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
        if (kvp < Sk) {
          kv8 = *(const bf16x8*)&K[kbase + (long long)kvp * D + c8 * 8];
          vv8 = *(const bf16x8*)&V[kbase + (long long)kvp * D + c8 * 8];
        }
        *(bf16x8*)&Ks[row][c8 * 8] = kv8;
#pragma unroll
        for (int j = 0; j < 8; ++j) Vts[c8 * 8 + j][row] = vv8[j];
      }
    }
    __syncthreads();

    // ---- S = scale * Q K^T  (per-wave, 2 m-tiles x 2 n-tiles) ----
    f32x4 s[2][2];
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        s[mt][nt] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kc = 0; kc < DCH; ++kc) {
          bf16x8 kfrag = *(const bf16x8*)&Ks[nt * 16 + lr][kc * 32 + lg * 8];
          s[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qf[mt][kc], kfrag, s[mt][nt], 0, 0, 0);
        }
      }

    // ---- mask + online softmax ----
#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) {
        int col = kb * KVBLK + nt * 16 + lr;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = q0 + mt * 16 + lg * 4 + r;
          bool dead = (col >= Sk) || (CAUSAL && col > row + causal_off);
          s[mt][nt][r] = dead ? -1e30f : s[mt][nt][r] * scale;
        }
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float smax = fmaxf(group16_max(s[mt][0][r]), group16_max(s[mt][1][r]));
        float m_new = fmaxf(m_run[mt][r], smax);
        float rescale = __expf(m_run[mt][r] - m_new);
        float p0 = __expf(s[mt][0][r] - m_new);
        float p1 = __expf(s[mt][1][r] - m_new);
        s[mt][0][r] = p0;
        s[mt][1][r] = p1;
        l_run[mt][r] = l_run[mt][r] * rescale + group16_sum(p0) + group16_sum(p1);
        m_run[mt][r] = m_new;
#pragma unroll
        for (int dt = 0; dt < DT; ++dt) acc[mt][dt][r] *= rescale;
      }
      // P tile (bf16) → per-wave LDS, becomes the next MFMA's A operand
#pragma unroll
      for (int nt = 0; nt < 2; ++nt)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          Ps[w][mt * 16 + lg * 4 + r][nt * 16 + lr] = f2bf(s[mt][nt][r]);
    }

    // ---- O += P V ----
#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
      bf16x8 pfrag = *(const bf16x8*)&Ps[w][mt * 16 + lr][lg * 8];
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        bf16x8 vfrag = *(const bf16x8*)&Vts[dt * 16 + lr][lg * 8];
        acc[mt][dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pfrag, vfrag, acc[mt][dt], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O = acc / l ----
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = q0 + mt * 16 + lg * 4 + r;
      if (row >= Sq) continue;
      float inv = l_run[mt][r] > 0.f ? 1.f / l_run[mt][r] : 0.f;
#pragma unroll
      for (int dt = 0; dt < DT; ++dt)
        O[qbase + (long long)row * D + dt * 16 + lr] = f2bf(acc[mt][dt][r] * inv);
    }
  }
}

extern "C" void fa_fwd_bf16(const void* q, const void* k, const void* v, void* o,
                            int B, int Hq, int Hkv, int Sq, int Sk, int D,
                            float scale, int causal, hipStream_t stream) {
  dim3 grid((Sq + QBLK - 1) / QBLK, Hq, B);
  dim3 block(NWAVES * WAVE);
  const short* Qp = (const short*)q;
  const short* Kp = (const short*)k;
  const short* Vp = (const short*)v;
  short* Op = (short*)o;
#define LAUNCH(DD, CC)                                                        \
  hipLaunchKernelGGL((fa_fwd_kernel<DD, CC>), grid, block, 0, stream, Qp, Kp, \
                     Vp, Op, B, Hq, Hkv, Sq, Sk, scale)
  if (D == 64) {
    if (causal) LAUNCH(64, true); else LAUNCH(64, false);
  } else if (D == 128) {
    if (causal) LAUNCH(128, true); else LAUNCH(128, false);
  } else {
    fprintf(stderr, "fa_fwd_bf16: unsupported head_dim %d (need 64 or 128)\n", D);
    abort();
  }
#undef LAUNCH
}
