// Flash-attention forward v2, bf16, gfx950 (CDNA4 MFMA 16x16x32).
//
// Serves: K1 diffusion self/cross-attention (D=64), K5
// Whisper encoder attention, K7 LLM prefill (D=128, causal, GQA).
//
// Structure (one workgroup = 4 waves, each wave owns 32 query rows):
//   per 64-wide KV block:
//     stage K and V row-major into LDS with vector 16B writes (no transpose
//     pass) using an XOR row swizzle to spread the 16-lane fragment reads
//     across banks (guide §6 G4),
//     QK^T: A=Q (registers), B=K (b128 LDS reads, hoisted across m-tiles),
//     online softmax in the MFMA C-layout (16-lane-group shuffle reductions),
//     P staged through per-wave LDS,
//     PV computed TRANSPOSED — O^T[d][q] = V^T·P^T — so the V operand reads
//     row-major V directly (strided u16) and the P operand reads row-major P
//     contiguously; no V^T staging exists at all.  Per-row softmax factors
//     reach the transposed accumulator via a 4-shuffle lane broadcast.
//
// MFMA fragment layouts (cdna_hip_programming.md §3):
//   A[l&15][(l>>4)*8+j], B[(l>>4)*8+j][l&15], C[(l>>4)*4+r][l&15].
#include "common.h"

#include <cstdio>

#define QROWS_PER_WAVE 32
#define KVBLK 64
#define NWAVES 4
#define QBLK (QROWS_PER_WAVE * NWAVES)  // 128 query rows per workgroup
#define PPAD 8

// XOR row swizzle for the shared K/V tiles: flips 16B units by kv bits 3..4,
// separating the four 16-lane groups' bank footprints. Applied identically on
// the staging writes and every read.
DEV_INLINE int kv_swz(int row, int byte_off) {
  return byte_off ^ (((row >> 3) & 3) << 4);
}

template <int D, bool CAUSAL>
__global__ __launch_bounds__(NWAVES * WAVE) void fa_fwd_kernel(
    const short* __restrict__ Q,  // [B, Hq, Sq, D] bf16
    const short* __restrict__ K,  // [B, Hkv, Sk, D]
    const short* __restrict__ V,  // [B, Hkv, Sk, D]
    short* __restrict__ O,        // [B, Hq, Sq, D]
    int B, int Hq, int Hkv, int Sq, int Sk, float scale) {
  constexpr int DCH = D / 32;   // QK^T k-chunks
  constexpr int DT = D / 16;    // d tiles
  constexpr int KROW = D + PPAD;  // LDS row pitch (halfwords)

  __shared__ alignas(16) short Ks[KVBLK][KROW];
  __shared__ alignas(16) short Vs[KVBLK][KROW];
  __shared__ alignas(16) short Ps[NWAVES][QROWS_PER_WAVE][KVBLK + PPAD];

  const int tid = threadIdx.x;
  const int w = tid / WAVE;
  const int l = tid % WAVE;
  const int lr = l & 15;
  const int lg = l >> 4;

  const int qblk = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const long long qbase = (((long long)b * Hq + h) * Sq) * D;
  const long long kbase = (((long long)b * Hkv + hkv) * Sk) * D;
  const int q0 = qblk * QBLK + w * QROWS_PER_WAVE;
  const int causal_off = Sk - Sq;

  // ---- Q fragments in registers ----
  bf16x8 qf[2][DCH];
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
    int qr = q0 + mt * 16 + lr;
#pragma unroll
    for (int kc = 0; kc < DCH; ++kc) {
      qf[mt][kc] = (qr < Sq)
          ? *(const bf16x8*)&Q[qbase + (long long)qr * D + kc * 32 + lg * 8]
          : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  // O^T accumulator: acc[dt][mt] rows = d (lg*4+r), cols = q (lr)
  f32x4 acc[DT][2];
  float m_run[2][4], l_run[2][4];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt)
#pragma unroll
    for (int mt = 0; mt < 2; ++mt) acc[dt][mt] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int mt = 0; mt < 2; ++mt)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_run[mt][r] = -1e30f;
      l_run[mt][r] = 0.f;
    }

  int nkb = (Sk + KVBLK - 1) / KVBLK;
  if (CAUSAL) {
    int max_kv = qblk * QBLK + QBLK - 1 + causal_off;
    int lim = (max_kv + KVBLK) / KVBLK;
    if (lim < nkb) nkb = lim;
  }

  for (int kb = 0; kb < nkb; ++kb) {
    __syncthreads();
    // ---- stage K and V (vector 16B writes, swizzled rows) ----
    {
      constexpr int CPR = D / 8;
      constexpr int NCH = KVBLK * CPR;
      for (int ci = tid; ci < NCH; ci += NWAVES * WAVE) {
        int row = ci / CPR, c8 = ci % CPR;
        int kvp = kb * KVBLK + row;
        bf16x8 kv8 = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
        bf16x8 vv8 = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
        if (kvp < Sk) {
          kv8 = *(const bf16x8*)&K[kbase + (long long)kvp * D + c8 * 8];
          vv8 = *(const bf16x8*)&V[kbase + (long long)kvp * D + c8 * 8];
        }
        int boff = kv_swz(row, c8 * 16);
        *(bf16x8*)((char*)&Ks[row][0] + boff) = kv8;
        *(bf16x8*)((char*)&Vs[row][0] + boff) = vv8;
      }
    }
    __syncthreads();

    // ---- S = scale * Q K^T : s[mt][nt], K fragments hoisted over mt ----
    f32x4 s[2][4];
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) s[mt][nt] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
      for (int kc = 0; kc < DCH; ++kc) {
        int krow = nt * 16 + lr;
        bf16x8 kfrag = *(const bf16x8*)(
            (char*)&Ks[krow][0] + kv_swz(krow, (kc * 32 + lg * 8) * 2));
        __builtin_amdgcn_s_setprio(1);
        s[0][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[0][kc], kfrag,
                                                           s[0][nt], 0, 0, 0);
        s[1][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[1][kc], kfrag,
                                                           s[1][nt], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }

    // ---- mask + online softmax (C layout: row q = lg*4+r, col kv = lr) ----
    float fac[2][4];  // exp rescale per (mt, r)
#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
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
        float smax = fmaxf(fmaxf(s[mt][0][r], s[mt][1][r]),
                           fmaxf(s[mt][2][r], s[mt][3][r]));
        smax = group16_max(smax);
        float m_new = fmaxf(m_run[mt][r], smax);
        float rs = __expf(m_run[mt][r] - m_new);
        float psum = 0.f;
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
          float p = __expf(s[mt][nt][r] - m_new);
          s[mt][nt][r] = p;
          psum += p;
        }
        l_run[mt][r] = l_run[mt][r] * rs + group16_sum(psum);
        m_run[mt][r] = m_new;
        fac[mt][r] = rs;
      }
      // P tile → per-wave LDS (row-major [32][KVBLK+8])
#pragma unroll
      for (int nt = 0; nt < 4; ++nt)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          Ps[w][mt * 16 + lg * 4 + r][nt * 16 + lr] = f2bf(s[mt][nt][r]);
    }

    // ---- rescale O^T: factor for column q = lr via 4-shuffle broadcast ----
#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
      int src = ((lr >> 2) << 4) | (l & 15);
      float f0 = __shfl(fac[mt][0], src, WAVE);
      float f1 = __shfl(fac[mt][1], src, WAVE);
      float f2 = __shfl(fac[mt][2], src, WAVE);
      float f3 = __shfl(fac[mt][3], src, WAVE);
      int rsel = lr & 3;
      float ft = rsel == 0 ? f0 : rsel == 1 ? f1 : rsel == 2 ? f2 : f3;
#pragma unroll
      for (int dt = 0; dt < DT; ++dt)
#pragma unroll
        for (int r = 0; r < 4; ++r) acc[dt][mt][r] *= ft;
    }

    // ---- O^T += V^T P^T : A = V^T (strided u16 reads), B = P^T (b128) ----
#pragma unroll
    for (int kc2 = 0; kc2 < 2; ++kc2) {
      bf16x8 pfrag[2];
#pragma unroll
      for (int mt = 0; mt < 2; ++mt)
        pfrag[mt] = *(const bf16x8*)&Ps[w][mt * 16 + lr][kc2 * 32 + lg * 8];
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        bf16x8 vfrag;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int vrow = kc2 * 32 + lg * 8 + j;
          vfrag[j] = *(const short*)(
              (char*)&Vs[vrow][0] + kv_swz(vrow, (dt * 16 + lr) * 2));
        }
        __builtin_amdgcn_s_setprio(1);
        acc[dt][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfrag, pfrag[0],
                                                             acc[dt][0], 0, 0, 0);
        acc[dt][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfrag, pfrag[1],
                                                             acc[dt][1], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }
  }

  // ---- epilogue: O[q][d] = O^T[d][q] / l[q] ----
  float inv[2];
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
    int src = ((lr >> 2) << 4) | (l & 15);
    float f0 = __shfl(l_run[mt][0], src, WAVE);
    float f1 = __shfl(l_run[mt][1], src, WAVE);
    float f2 = __shfl(l_run[mt][2], src, WAVE);
    float f3 = __shfl(l_run[mt][3], src, WAVE);
    int rsel = lr & 3;
    float lv = rsel == 0 ? f0 : rsel == 1 ? f1 : rsel == 2 ? f2 : f3;
    inv[mt] = lv > 0.f ? 1.f / lv : 0.f;
  }
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
    int row = q0 + mt * 16 + lr;  // q index (column of O^T)
    if (row >= Sq) continue;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        O[qbase + (long long)row * D + dt * 16 + lg * 4 + r] =
            f2bf(acc[dt][mt][r] * inv[mt]);
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
