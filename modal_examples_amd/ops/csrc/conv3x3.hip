// K3: hand-written NCHW implicit-GEMM 3x3 conv (stride 1, pad 1) on MFMA.
//
// Replaces MIOpen/CK for the SDXL VAE/UNet conv pyramid (reference trigger:
// 06_gpu_and_ml/stable_diffusion/text_to_image.py:114 VAE decode path,
// flux.py:259-261).  MIOpen's fastest solvers insert NCHW<->NHWC transpose
// pairs (~19 ms of the 390 ms SDXL step); this kernel consumes NCHW directly
// and does its one unavoidable transpose (x-contiguous -> c-contiguous) in
// LDS during staging.
//
// Formulation: 9 shifted GEMMs.  out[k][p] = sum_tap sum_c W[tap][k][c] *
// in[c][p+tap].  M = 64 out-channels, N = 256 pixels (8 rows x 32 cols),
// K-dim = C in chunks of 16 (mfma_f32_32x32x16_bf16).
//
//  - Input patch staged per 16-channel chunk as LDS [10 rows][34 cols][16 c]
//    (c innermost): the MFMA B-fragment read (8 consecutive c at the lane's
//    pixel) is one ds_read_b128 at 32 B col-stride -> 64 lanes cover a
//    contiguous span, every bank hit evenly (conflict-free; guide §6 G4
//    applies to same-column strides, not contiguous spans).
//  - Weights repacked on host to [9][Kpad][C16] (c contiguous) so the MFMA
//    A-fragment is one bf16x8 global read; per-k-tile weight working set
//    (64 x C x 9 x 2B <= 590 KB) stays L2-resident across the XCD-chunked
//    run of pixel tiles (bijective chunk swizzle, guide §5 T1/m204).
//  - Wave w owns output rows {2w, 2w+1} and BOTH 32-k fragments: 36 MFMA per
//    18 ds_read_b128 per chunk (2:1, the m97 GEMM ratio).
//  - Epilogue fuses bias add and an optional residual add (VAEResnet skip).
//
// The kernel is stride-1/pad-1 only; stride-2 downsample convs (3 per UNet
// fwd) stay on the library path.
#include "common.h"

#include <cstdio>
#include <cstdlib>

#define CV_NW 4      // waves per block
#define CV_COLS 32   // output cols per block
#define CV_BK 64     // output channels per block
#define CV_CC 16     // input-channel chunk (MFMA K)

typedef __attribute__((ext_vector_type(16))) float f32x16c;

// RPW = output rows per wave (block rows = CV_NW*RPW).  RPW=2 runs 3
// waves/SIMD; RPW=4 doubles MFMA per staged byte/A-read at 2 waves/SIMD.
// LDS patch: [rows+2][CV_COLS+2][CV_CC] bf16, c innermost.
#define PATCH_C (CV_COLS + 2)

// weight tile in LDS: [9 taps][CV_BK k][CV_CC c], staged once per c-chunk and
// shared by all 4 waves (per-wave global A-reads were 4x-redundant L2
// traffic — the v1 bottleneck).
#define WLDS_ELEMS (9 * CV_BK * CV_CC)
#define WSLOTS (WLDS_ELEMS / 8)  // bf16x8 units
#define WSLOTS_PER_T ((WSLOTS + CV_NW * WAVE - 1) / (CV_NW * WAVE))

// UP: fused nearest-2x upsample — the conv reads the half-res source
// directly (out pixel (y,x) <- src[y/2][x/2]), eliminating the materialized
// F.interpolate pass before every Upsample conv.  H/W are OUTPUT dims;
// Hs/Ws the source's.
// GN: fuse GroupNorm+SiLU into the staging read — silu(x*scale[c]+shift[c])
// with per-(n,c) coefficients from gn_conv_coeffs_bf16 (norms.hip).  The
// gn_norm write+read pass over the full tensor disappears.
template <bool UP, bool GN, int RPW>
__global__ __launch_bounds__(CV_NW * WAVE) void conv3x3_kernel(
    const short* __restrict__ in, const short* __restrict__ wr,
    const float* __restrict__ bias, const short* __restrict__ res,
    short* __restrict__ out, const float* __restrict__ gn_scale,
    const float* __restrict__ gn_shift, int C, int H, int W, int Hs, int Ws,
    int K, int C16, int Kpad, int npix_x, int npix, int nk) {
  constexpr int CV_ROWS = CV_NW * RPW;
  constexpr int PATCH_R = CV_ROWS + 2;
  constexpr int PATCH_ELEMS = PATCH_R * PATCH_C * CV_CC;
  constexpr int STAGE_SLOTS = PATCH_R * PATCH_C * 2;  // (row, col, c-oct)
  constexpr int SLOTS_PER_T =
      (STAGE_SLOTS + CV_NW * WAVE - 1) / (CV_NW * WAVE);
  __shared__ alignas(16) short patch[PATCH_ELEMS];
  __shared__ alignas(16) short wlds[WLDS_ELEMS];

  const int tid = threadIdx.x;
  const int w = tid / WAVE;
  const int l = tid % WAVE;
  const int l31 = l & 31;
  const int hi5 = l >> 5;
  const int n = blockIdx.y;

  // bijective XCD-chunk swizzle (m204): each XCD runs a contiguous range of
  // block ids = consecutive pixel tiles of ONE k-tile -> weights L2-hit.
  int bid = blockIdx.x;
  {
    int nwg = npix * nk;
    int q = nwg >> 3, r = nwg & 7;
    int xcd = bid & 7, pos = bid >> 3;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int ktile = bid / npix;
  const int pix = bid - ktile * npix;
  const int y0 = (pix / npix_x) * CV_ROWS;
  const int x0 = (pix - (pix / npix_x) * npix_x) * CV_COLS;
  const int k0 = ktile * CV_BK;

  const long long in_n = (long long)n * C * Hs * Ws;
  const int nc = C16 / CV_CC;

  // ---- staging: slot s = (row, col, c-oct); thread gathers 8 strided
  // channel values (coalesced across lanes: consecutive threads read
  // consecutive x) and writes ONE bf16x8 to the c-contiguous LDS slot.
  bf16x8 sreg[SLOTS_PER_T];
  auto stage_load = [&](int cc) {
    const int c0 = cc * CV_CC;
#pragma unroll
    for (int i = 0; i < SLOTS_PER_T; ++i) {
      int s = i * CV_NW * WAVE + tid;
      bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
      if (s < STAGE_SLOTS) {
        int col = s % PATCH_C;
        int u = s / PATCH_C;
        int oct = u & 1;
        int row = u >> 1;
        int y = y0 + row - 1;
        int x = x0 + col - 1;
        if (y >= 0 && y < H && x >= 0 && x < W) {
          const int sy = UP ? (y >> 1) : y;
          const int sx = UP ? (x >> 1) : x;
          const long long hw = (long long)Hs * Ws;
          const short* src = in + in_n + (long long)(c0 + oct * 8) * hw +
                             (long long)sy * Ws + sx;
          // GN coeffs: tiny [N, C16] arrays, L1/L2-broadcast across pixels
          const float* gsc = GN ? gn_scale + (long long)n * C16 + c0 + oct * 8
                                : nullptr;
          const float* gsh = GN ? gn_shift + (long long)n * C16 + c0 + oct * 8
                                : nullptr;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            // keep the load as a SELECT (not a branch): branchy loads
            // serialize behind exec-mask updates and exposed the staging
            // latency (measured 13.3 -> 9.8 img/s whole-step)
            int c = c0 + oct * 8 + j;
            short raw = (c < C) ? src[(long long)j * hw] : (short)0;
            if (GN) {
              float f = bf2f(raw) * gsc[j] + gsh[j];
              f = f / (1.f + __expf(-f));
              raw = (c < C) ? f2bf(f) : (short)0;
            }
            v[j] = raw;
          }
        }
      }
      sreg[i] = v;
    }
  };
  auto stage_write = [&]() {
#pragma unroll
    for (int i = 0; i < SLOTS_PER_T; ++i) {
      int s = i * CV_NW * WAVE + tid;
      if (s < STAGE_SLOTS) {
        int col = s % PATCH_C;
        int u = s / PATCH_C;
        int oct = u & 1;
        int row = u >> 1;
        *(bf16x8*)&patch[(row * PATCH_C + col) * CV_CC + oct * 8] = sreg[i];
      }
    }
  };

  // weight staging: slot = (tap, k-row, c-oct); one bf16x8 each.
  bf16x8 wreg[WSLOTS_PER_T];
  auto wstage_load = [&](int cc) {
    const int c0 = cc * CV_CC;
#pragma unroll
    for (int i = 0; i < WSLOTS_PER_T; ++i) {
      int s = i * CV_NW * WAVE + tid;
      if (s < WSLOTS) {
        int oct = s & 1;
        int kr = (s >> 1) & (CV_BK - 1);
        int tap = (s >> 1) >> 6;  // 64 k-rows per tap
        wreg[i] = *(const bf16x8*)&wr[((long long)tap * Kpad + k0 + kr) * C16 +
                                      c0 + oct * 8];
      }
    }
  };
  auto wstage_write = [&]() {
#pragma unroll
    for (int i = 0; i < WSLOTS_PER_T; ++i) {
      int s = i * CV_NW * WAVE + tid;
      if (s < WSLOTS) {
        *(bf16x8*)&wlds[s * 8] = wreg[i];
      }
    }
  };

  // accumulators: [wave-row rr][k-fragment mf]
  f32x16c acc[RPW][2];
#pragma unroll
  for (int a = 0; a < RPW; ++a)
#pragma unroll
    for (int b = 0; b < 2; ++b)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[a][b][r] = 0.f;

  stage_load(0);
  wstage_load(0);
  for (int cc = 0; cc < nc; ++cc) {
    __syncthreads();  // previous compute done; LDS free
    stage_write();
    wstage_write();
    __syncthreads();  // patch + weights ready
    if (cc + 1 < nc) {
      stage_load(cc + 1);
      wstage_load(cc + 1);
    }

#pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      const int dy = tap / 3, dx = tap - 3 * (tap / 3);
      bf16x8 a0 = *(const bf16x8*)&wlds[(tap * CV_BK + l31) * CV_CC + hi5 * 8];
      bf16x8 a1 = *(const bf16x8*)&wlds[(tap * CV_BK + 32 + l31) * CV_CC + hi5 * 8];
#pragma unroll
      for (int rr = 0; rr < RPW; ++rr) {
        const int row = w * RPW + rr;
        bf16x8 b = *(const bf16x8*)&patch[((row + dy) * PATCH_C + l31 + dx) *
                                              CV_CC + hi5 * 8];
        __builtin_amdgcn_s_setprio(1);
        acc[rr][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b, acc[rr][0], 0, 0, 0);
        acc[rr][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b, acc[rr][1], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }
  }

  // ---- epilogue: bias (+ residual) add, masked NCHW store.
  const int x = x0 + l31;
  if (x < W) {
#pragma unroll
    for (int rr = 0; rr < RPW; ++rr) {
      const int y = y0 + w * RPW + rr;
      if (y >= H) continue;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int k = k0 + mf * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi5;
          if (k < K) {
            const long long o = ((long long)n * K + k) * H * W +
                                (long long)y * W + x;
            float v = acc[rr][mf][r] + bias[k];
            if (res != nullptr) v += bf2f(res[o]);
            out[o] = f2bf(v);
          }
        }
      }
    }
  }
}

extern "C" void conv3x3_bf16(const void* in, const void* wrepack,
                             const void* bias, const void* residual, void* out,
                             const float* gn_scale, const float* gn_shift,
                             int N, int C, int H, int W, int K, int C16,
                             int Kpad, int upsample, hipStream_t stream) {
  // H/W are the OUTPUT dims; with upsample the source is H/2 x W/2.
  const int Hs = upsample ? H / 2 : H;
  const int Ws = upsample ? W / 2 : W;
  // rows-per-wave A/B (MODAL_AMD_CONV_RPW=4): doubles MFMA per staged
  // byte/A-read at 2 waves/SIMD occupancy.
  static const int rpw = [] {
    const char* e = getenv("MODAL_AMD_CONV_RPW");
    return (e && atoi(e) == 4) ? 4 : 2;
  }();
  const int rows = CV_NW * rpw;
  const int npix_x = (W + CV_COLS - 1) / CV_COLS;
  const int npix_y = (H + rows - 1) / rows;
  const int npix = npix_x * npix_y;
  const int nk = (K + CV_BK - 1) / CV_BK;
  dim3 grid(npix * nk, N);
  dim3 block(CV_NW * WAVE);
  const bool gn = gn_scale != nullptr;
#define CVL(UPV, GNV, RPWV) \
  hipLaunchKernelGGL((conv3x3_kernel<UPV, GNV, RPWV>), grid, block, 0, \
                     stream, (const short*)in, (const short*)wrepack, \
                     (const float*)bias, (const short*)residual, (short*)out, \
                     gn_scale, gn_shift, C, H, W, Hs, Ws, K, C16, Kpad, \
                     npix_x, npix, nk)
  if (rpw == 4) {
    if (upsample) {
      if (gn) CVL(true, true, 4); else CVL(true, false, 4);
    } else {
      if (gn) CVL(false, true, 4); else CVL(false, false, 4);
    }
  } else {
    if (upsample) {
      if (gn) CVL(true, true, 2); else CVL(true, false, 2);
    } else {
      if (gn) CVL(false, true, 2); else CVL(false, false, 2);
    }
  }
#undef CVL
}
