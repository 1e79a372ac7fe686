// Flash-attention forward v7 — 32x32x16 MFMA with LANE-LOCAL online softmax.
//
// The v3-v6 kernels (attention_fwd.hip) are softmax-LATENCY-bound: ablation
// measured the 16-lane-group shuffle/exp chains at ~56% of kernel time.  This
// structure eliminates cross-lane reductions almost entirely (the m214-style
// swapped-operand recipe, re-derived for gfx950):
//
//   QK^T is computed TRANSPOSED with mfma_f32_32x32x16_bf16:
//     S^T = K · Q^T  →  each lane holds 16 of the 32 kv-scores for ONE query
//     (C layout col = q = lane&31); row-softmax becomes 15 in-register ops +
//     ONE shfl_xor(32) to combine with the partner lane.  Rescale factors are
//     lane-local (no broadcast).
//   P converts to bf16 B-fragments IN REGISTERS: v_cvt_pk_bf16_f32 pairs +
//     permlane32_swap half-exchanges (semantics measured: result pair is
//     (A_lo||B_lo , A_hi||B_hi)) — P never touches LDS.
//   PV: O^T = V^T · P^T, V^T fragments via ds_read_b64_tr_b16 hardware
//     transpose reads (4x4 lane-grid exchange, measured in probe_tr16.cpp).
//
// Fragment layouts (verified on-device, scripts/probe_mfma32.cpp):
//   A[m][k]: lane holds A[l&31][(l>>5)*8+j];  B[k][n]: B[(l>>5)*8+j][l&31]
//   C[m][n]: lane holds C[(reg&3)+8*(reg>>2)+4*(l>>5)][l&31], reg 0..15.
#include "common.h"

#include <cstdio>
#include <cstdlib>

#define FA32_NWAVES 4
#define FA32_KVBLK 32
#define FA32_PPAD 8

typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) short bf16x4_v7;
typedef __attribute__((address_space(3))) bf16x4_v7* lds_tr_ptr7;
typedef __attribute__((ext_vector_type(2))) int i32x2;

DEV_INLINE int kv_swz7(int row, int byte_off) {
  return byte_off ^ (((row >> 3) & 3) << 4);
}

DEV_INLINE unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

struct FA32Strides {
  long long qb, qh, qs, kb, kh, ks, vb, vh, vs, ob, oh, os;
};

// KVB: kv rows per block iteration.  64 halves the barrier/softmax rounds
// per token vs the r1 kernel's 32 (the m214-ladder "KVBLK=64" lever).
// DEFER: defer-max rescale skipping (HK THR=8) — keep the old running max
// while per-block growth <= 8, so P = exp(s - m_old) <= e^8 (f32 acc safe)
// and the O-wide rescale is skipped entirely.
template <int D, bool CAUSAL, int KVB, bool DEFER>
__global__ __launch_bounds__(FA32_NWAVES * WAVE) void fa32_kernel(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, short* __restrict__ O,
    int B, int Hq, int Hkv, int Sq, int Sk, float scale, FA32Strides st) {
  constexpr int KROW = D + FA32_PPAD;
  constexpr int DCH = D / 16;  // QK^T k-chunks (K-dim 16)
  constexpr int DT = D / 32;   // PV d-tiles (M-dim 32)
  constexpr int QBLK = 32 * FA32_NWAVES;
  constexpr int KF = KVB / 32;  // 32-kv score fragments per iteration

  __shared__ alignas(16) short Ks[KVB][KROW];
  __shared__ alignas(16) short Vs[KVB][KROW];

  const int tid = threadIdx.x;
  const int w = tid / WAVE;
  const int l = tid % WAVE;
  const int l31 = l & 31;
  const int hi5 = l >> 5;  // 0 for lanes 0-31, 1 for 32-63

  const int qblk = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);

  const long long qoff = b * st.qb + h * st.qh;
  const long long koff = b * st.kb + hkv * st.kh;
  const long long voff = b * st.vb + hkv * st.vh;
  const long long ooff = b * st.ob + h * st.oh;
  const int q0 = qblk * QBLK + w * 32;
  const int my_q = q0 + l31;  // this lane's query row
  const int causal_off = Sk - Sq;

  // Q^T B-fragments: qreg[c][j] = Q[my_q][c*16 + hi5*8 + j].
  // D=128 would spend 32 VGPR on Q (dropping occupancy to 2 waves/SIMD);
  // since Q is re-read once per KV block and stays L2-resident, the large-D
  // path reloads fragments from global inside the QK loop instead.
  // measured: reloading Q from L2 regressed D=128 22-30% (occupancy stayed
  // at 2 waves; the extra global traffic was pure cost) — registers for all D
  constexpr bool QREG = true;
  bf16x8 qreg[QREG ? DCH : 1];
  const long long qrow_off = qoff + (long long)my_q * st.qs;
  const bool q_ok = my_q < Sq;
  if (QREG) {
#pragma unroll
    for (int c = 0; c < (QREG ? DCH : 1); ++c) {
      qreg[c] = q_ok
          ? *(const bf16x8*)&Q[qrow_off + c * 16 + hi5 * 8]
          : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  f32x16 acc[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc[dt][r] = 0.f;
  float m_run = -1e30f, l_run = 0.f;

  int nkb = (Sk + KVB - 1) / KVB;
  if (CAUSAL) {
    int max_kv = qblk * QBLK + QBLK - 1 + causal_off;
    int lim = (max_kv + KVB) / KVB;
    if (lim < nkb) nkb = lim;
  }

  // async register staging: next block loads issue before this block computes
  constexpr int CPR = D / 8;
  constexpr int CPT = (KVB * CPR) / (FA32_NWAVES * WAVE);
  bf16x8 kreg[CPT], vreg[CPT];
  auto load_chunks = [&](int kb) {
#pragma unroll
    for (int i = 0; i < CPT; ++i) {
      int ci = i * FA32_NWAVES * WAVE + tid;
      int row = ci / CPR, c8 = ci % CPR;
      int kvp = kb * KVB + row;
      if (kvp < Sk) {
        kreg[i] = *(const bf16x8*)&K[koff + kvp * st.ks + c8 * 8];
        vreg[i] = *(const bf16x8*)&V[voff + kvp * st.vs + c8 * 8];
      } else {
        kreg[i] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
        vreg[i] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  };
  auto write_chunks = [&]() {
#pragma unroll
    for (int i = 0; i < CPT; ++i) {
      int ci = i * FA32_NWAVES * WAVE + tid;
      int row = ci / CPR, c8 = ci % CPR;
      int boff = kv_swz7(row, c8 * 16);
      *(bf16x8*)((char*)&Ks[row][0] + boff) = kreg[i];
      *(bf16x8*)((char*)&Vs[row][0] + boff) = vreg[i];
    }
  };

  load_chunks(0);
  for (int kb = 0; kb < nkb; ++kb) {
    __syncthreads();
    write_chunks();
    __syncthreads();
    if (kb + 1 < nkb) load_chunks(kb + 1);

    // ---- S^T = K · Q^T (KF fragments of 32 kv each) ----
    f32x16 s[KF];
#pragma unroll
    for (int f = 0; f < KF; ++f)
#pragma unroll
      for (int r = 0; r < 16; ++r) s[f][r] = 0.f;
#pragma unroll
    for (int f = 0; f < KF; ++f) {
#pragma unroll
      for (int c = 0; c < DCH; ++c) {
        // A = K[kv][d]: lane holds K[f*32 + l31][c*16 + hi5*8 + j]
        int krow = f * 32 + l31;
        bf16x8 kfrag = *(const bf16x8*)(
            (char*)&Ks[krow][0] + kv_swz7(krow, (c * 16 + hi5 * 8) * 2));
        bf16x8 qf;
        if (QREG) {
          qf = qreg[c % (QREG ? DCH : 1)];
        } else {
          qf = q_ok ? *(const bf16x8*)&Q[qrow_off + c * 16 + hi5 * 8]
                    : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
        }
        __builtin_amdgcn_s_setprio(1);
        s[f] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfrag, qf, s[f], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }

    // ---- mask + lane-local online softmax (this lane owns query my_q) ----
    bool full = (kb * KVB + KVB <= Sk) &&
                (!CAUSAL || kb * KVB + KVB - 1 <= my_q + causal_off);
#pragma unroll
    for (int f = 0; f < KF; ++f)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int kvp = kb * KVB + f * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi5;
        bool dead = !full && ((kvp >= Sk) || (CAUSAL && kvp > my_q + causal_off));
        s[f][r] = dead ? -1e30f : s[f][r] * scale;
      }
    float smax = s[0][0];
#pragma unroll
    for (int f = 0; f < KF; ++f)
#pragma unroll
      for (int r = 0; r < 16; ++r) smax = fmaxf(smax, s[f][r]);
    smax = fmaxf(smax, __shfl_xor(smax, 32, WAVE));  // partner combine
    if (!DEFER || kb == 0 ||
        __ballot(smax > m_run + 8.f) != 0ull) {
      float m_new = fmaxf(m_run, smax);
      float rs = __expf(m_run - m_new);
      if (kb > 0 && __ballot(rs < 0.999999f) != 0ull) {
        l_run *= rs;
#pragma unroll
        for (int dt = 0; dt < DT; ++dt)
#pragma unroll
          for (int r = 0; r < 16; ++r) acc[dt][r] *= rs;
      }
      m_run = m_new;
    }
    float psum = 0.f;
#pragma unroll
    for (int f = 0; f < KF; ++f)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        s[f][r] = __expf(s[f][r] - m_run);
        psum += s[f][r];
      }
    psum += __shfl_xor(psum, 32, WAVE);
    l_run = (kb == 0) ? psum : l_run + psum;

    // ---- P → bf16 B-fragments in registers (cvt_pk + permlane32_swap) ----
    // own regs pack kv pairs; half-exchange composes the 16-kv chunks:
    //   chunk c frag words = [A'(pk01,pk45), A'(pk23,pk67), B'(same), B'(same)]
    bf16x8 pfrag[2 * KF];
#pragma unroll
    for (int c = 0; c < 2 * KF; ++c) {
      float* sv = (float*)&s[c >> 1] + (c & 1) * 8;
      int pk01 = (int)cvt_pk_bf16(sv[0], sv[1]);
      int pk23 = (int)cvt_pk_bf16(sv[2], sv[3]);
      int pk45 = (int)cvt_pk_bf16(sv[4], sv[5]);
      int pk67 = (int)cvt_pk_bf16(sv[6], sv[7]);
      i32x2 x = __builtin_amdgcn_permlane32_swap(pk01, pk45, false, false);
      i32x2 y = __builtin_amdgcn_permlane32_swap(pk23, pk67, false, false);
      union {
        int w[4];
        bf16x8 v;
      } u;
      u.w[0] = x[0];
      u.w[1] = y[0];
      u.w[2] = x[1];
      u.w[3] = y[1];
      pfrag[c] = u.v;
    }

    // ---- O^T += V^T · P^T (V^T via tr16 hardware transpose reads) ----
#pragma unroll
    for (int c = 0; c < 2 * KF; ++c) {
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        // lane (within its 16-group: 4a+b) issues the source read for
        // V[c*16 + hi5*8 + (l&15)>>2 .. +4][dt*32 + ((l>>4)&1)*16 + 4a]
        int row0 = c * 16 + hi5 * 8 + ((l & 15) >> 2);
        int dbase = dt * 32 + ((l >> 4) & 1) * 16 + ((l & 3) * 4);
        bf16x4_v7 lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds_tr_ptr7)(
            (char*)&Vs[0][0] + row0 * (KROW * 2) + kv_swz7(row0, dbase * 2)));
        bf16x4_v7 hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds_tr_ptr7)(
            (char*)&Vs[0][0] + (row0 + 4) * (KROW * 2) +
            kv_swz7(row0 + 4, dbase * 2)));
        bf16x8 vfrag;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          vfrag[j] = lo[j];
          vfrag[j + 4] = hi[j];
        }
        __builtin_amdgcn_s_setprio(1);
        acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfrag, pfrag[c],
                                                          acc[dt], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }
  }

  // ---- epilogue: O[q][d] = O^T[d][q] / l ----
  if (my_q < Sq) {
    float inv = l_run > 0.f ? 1.f / l_run : 0.f;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int d = dt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi5;
        O[ooff + (long long)my_q * st.os + d] = f2bf(acc[dt][r] * inv);
      }
  }
}

extern "C" void fa32_fwd_strided_bf16(
    const void* q, const void* k, const void* v, void* o, int B, int Hq,
    int Hkv, int Sq, int Sk, int D, float scale, int causal,
    const long long* strides, hipStream_t stream) {
  const short* Qp = (const short*)q;
  const short* Kp = (const short*)k;
  const short* Vp = (const short*)v;
  short* Op = (short*)o;
  FA32Strides st;
  st.qb = strides[0]; st.qh = strides[1]; st.qs = strides[2];
  st.kb = strides[3]; st.kh = strides[4]; st.ks = strides[5];
  st.vb = strides[6]; st.vh = strides[7]; st.vs = strides[8];
  st.ob = strides[9]; st.oh = strides[10]; st.os = strides[11];
  dim3 grid((Sq + 127) / 128, Hq, B);
  dim3 block(FA32_NWAVES * WAVE);
  // KVB is chosen per head-dim (A/B-measured, profiles/attn_ab_r2.txt):
  //   D=64:  KVB=64 wins (447 vs 407 TF — fewer barrier/softmax rounds)
  //   D=128: KVB=64 LOST 40% (causal instantiation lands at 1 wave/SIMD —
  //          the r1 double-buffer failure mode); only KVB=32 is built.
  // MODAL_AMD_FA_NODEFER disables defer-max rescale skipping (+5-10% A/B).
  static const bool defer_env = getenv("MODAL_AMD_FA_NODEFER") == nullptr;
#define L32K(DD, CC, KK)                                                      \
  do {                                                                        \
    if (defer_env)                                                            \
      hipLaunchKernelGGL((fa32_kernel<DD, CC, KK, true>), grid, block, 0,     \
                         stream, Qp, Kp, Vp, Op, B, Hq, Hkv, Sq, Sk, scale, st); \
    else                                                                      \
      hipLaunchKernelGGL((fa32_kernel<DD, CC, KK, false>), grid, block, 0,    \
                         stream, Qp, Kp, Vp, Op, B, Hq, Hkv, Sq, Sk, scale, st); \
  } while (0)
  if (D == 64) {
    if (causal) L32K(64, true, 64); else L32K(64, false, 64);
  } else if (D == 128) {
    if (causal) L32K(128, true, 32); else L32K(128, false, 32);
  } else {
    fprintf(stderr, "fa32: unsupported head_dim %d\n", D);
    abort();
  }
#undef L32K
}
