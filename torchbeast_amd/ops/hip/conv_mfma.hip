// MFMA implicit-GEMM conv trunk for AtariNet on gfx950 (CDNA4).
//
// Replaces the MIOpen/aten conv path for the learner batch (any N) with
// hand-written bf16 matrix-core kernels. Design (MI355X-first, not a port —
// the reference torchbeast/monobeast.py:552-559 uses stock nn.Conv2d):
//
// * Every conv layer is an implicit GEMM on v_mfma_f32_16x16x32_bf16:
//   M = output positions, N = output channels, K = KH*KW*C_in. The im2col
//   matrix is never materialized; A-fragments are gathered from an LDS
//   copy of the input tile.
// * Intermediate activations are NHWC bf16 so both the A-gather and the
//   epilogue writes are contiguous along channels; conv1 keeps the u8
//   frames NCHW (C=4 is too thin for NHWC runs) and fuses the /255
//   normalization into LDS staging.
// * Each workgroup owns a sample group (or an output-row band) and stages
//   its whole input tile in LDS ONCE: the K-loop has no barriers at all.
//   Weight fragments stream straight from global memory (L2-resident:
//   16-72 KB per layer) so no LDS double-buffering is needed.
// * LDS layouts use the XOR swizzle byte^=((byte>>7)&7)<<4 where the
//   A-fragment stride is a multiple of 128 B (conv2/3: stride*C_in*2B =
//   128 B would otherwise be a 16-way bank conflict).
// * dgrad reuses the same kernel template: stride-1 correlation over a
//   zero-padded (conv3) or zero-dilated (conv2, stride 2) dY tile staged
//   in LDS, with rotated/permuted weights prepared host-side and the
//   ReLU mask fused into the epilogue.
// * wgrad is a transposed GEMM per ky-slice: dW[n][ky][kx*C+c] =
//   sum_m dY^T[n][m] * X[m][kx*C+c], reduced over M in fp32 MFMA
//   accumulators with per-chunk partials (no atomics) + a reduce kernel.
//
// Numerics: bf16 operands, fp32 accumulation/bias/ReLU. Oracles:
// tests/test_conv_mfma.py compares against fp32 F.conv2d.

#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <vector>

#include "conv_mfma.h"

namespace tbamd {

namespace {

using bf16x4 = __attribute__((ext_vector_type(4))) __bf16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int kThreads = 512;  // 8 waves.
constexpr int kWaves = kThreads / 64;

__device__ __forceinline__ int swz(int byte) {
  // Spread 128 B-strided accesses over eight 16 B slots (CDNA4 guide §6 G4).
  return byte ^ (((byte >> 7) & 7) << 4);
}

__device__ __forceinline__ bf16x8 lds_read8_swz(const char* smem, int elem) {
  return *reinterpret_cast<const bf16x8*>(smem + swz(elem * 2));
}

__device__ __forceinline__ void lds_write8_swz(char* smem, int elem, bf16x8 v) {
  *reinterpret_cast<bf16x8*>(smem + swz(elem * 2)) = v;
}

// ---------------------------------------------------------------------------
// conv1: u8 NCHW frames -> NHWC bf16, k-order (c, ky, kx) (the native
// PyTorch weight flatten), /255 fused into staging. No swizzle needed:
// the A-fragment stride is ST*2B = 8 B (2-way conflicts are free).
// ---------------------------------------------------------------------------

template <int CI, int IH, int IW, int KH, int KW, int ST, int CO, int OH,
          int OW, int OYT = OH>
__global__ __launch_bounds__(kThreads) void conv1_u8_kernel(
    const uint8_t* __restrict__ in,  // [N, CI, IH, IW]
    const __bf16* __restrict__ W,    // [CO, CI*KH*KW] (c,ky,kx)-major
    const float* __restrict__ bias,  // [CO]
    __bf16* __restrict__ out,        // [N, OH, OW, CO]
    int N) {
  constexpr int IWP = (IW + 7) & ~7;  // pad rows to 8 elements (16 B)
  constexpr int K = CI * KH * KW;
  constexpr int LROWS = (OYT - 1) * ST + KH;  // staged input rows per band
  constexpr int M_BLK = OYT * OW;
  constexpr int MF = (M_BLK + 15) / 16;
  constexpr int NF = CO / 16;
  constexpr int MAX_MF = (MF + kWaves - 1) / kWaves;
  constexpr int BANDS = (OH + OYT - 1) / OYT;
  static_assert(KW == 8, "conv1 kernel assumes KW == 8 fragment runs");

  extern __shared__ char smem[];  // CI * LROWS * IWP bf16
  __bf16* img = reinterpret_cast<__bf16*>(smem);

  const int s = blockIdx.x / BANDS;
  const int band = blockIdx.x % BANDS;
  const int oy0 = band * OYT;
  const int ly0 = oy0 * ST;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int ln = lane & 15;
  const int lg = lane >> 4;

  // Stage this band's input rows, converting u8 -> bf16/255. One thread
  // per 16-px chunk (84-px rows = 5 full chunks + a 4-px tail).
  {
    constexpr int CHUNKS_PER_ROW = (IW + 15) / 16;
    constexpr int NCHUNK = CI * LROWS * CHUNKS_PER_ROW;
    const uint8_t* base = in + (int64_t)s * CI * IH * IW;
    for (int idx = tid; idx < NCHUNK; idx += kThreads) {
      const int c = idx / (LROWS * CHUNKS_PER_ROW);
      const int rem = idx % (LROWS * CHUNKS_PER_ROW);
      const int lrow = rem / CHUNKS_PER_ROW;
      const int row = ly0 + lrow;
      const int ch = rem % CHUNKS_PER_ROW;
      const int x0 = ch * 16;
      const int npx = (row < IH) ? min(16, IW - x0) : 0;
      const uint8_t* src = base + (c * IH + row) * IW + x0;
      __bf16* dst = &img[(c * LROWS + lrow) * IWP + x0];
      constexpr float kInv = 1.0f / 255.0f;
      if (npx == 16) {
        // Row starts are only 4 B-aligned (84 px rows): four u32 loads.
        const uint32_t* s32 = reinterpret_cast<const uint32_t*>(src);
        uint32_t w[4];
#pragma unroll
        for (int q = 0; q < 4; ++q) w[q] = s32[q];
        bf16x8 lo, hi;
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          lo[i] = (__bf16)(kInv * ((w[i >> 2] >> (8 * (i & 3))) & 0xffu));
          hi[i] = (__bf16)(kInv * ((w[2 + (i >> 2)] >> (8 * (i & 3))) & 0xffu));
        }
        *reinterpret_cast<bf16x8*>(dst) = lo;
        *reinterpret_cast<bf16x8*>(dst + 8) = hi;
      } else {
        for (int i = 0; i < npx; ++i) dst[i] = (__bf16)(kInv * src[i]);
        for (int i = npx; i < 16 && x0 + i < IWP; ++i) dst[i] = (__bf16)0.f;
      }
    }
    // Zero the row pad (A-fragments of the right edge read into it).
    for (int idx = tid; idx < CI * LROWS * (IWP - IW); idx += kThreads) {
      const int r = idx / (IWP - IW);
      img[r * IWP + IW + idx % (IWP - IW)] = (__bf16)0.f;
    }
  }
  __syncthreads();

  // Per-wave fragment bookkeeping.
  int aoff[MAX_MF];  // LDS element offset of the band-local (oy*ST) row
  int nmf = 0;
  for (int f = wave; f < MF; f += kWaves, ++nmf) {
    const int m = min(f * 16 + ln, M_BLK - 1);
    const int oy = m / OW, ox = m % OW;
    aoff[nmf] = (oy * ST) * IWP + ox * ST;
  }
  const int mvalid = min(OYT, OH - oy0) * OW;

  f32x4 acc[MAX_MF][NF];
#pragma unroll
  for (int f = 0; f < MAX_MF; ++f)
#pragma unroll
    for (int j = 0; j < NF; ++j) acc[f][j] = {0.f, 0.f, 0.f, 0.f};

  const int kfrag = (lane >> 4) * 8;  // this lane group's K offset
#pragma unroll
  for (int ks = 0; ks < K / 32; ++ks) {
    const int k = ks * 32 + kfrag;
    const int c = k >> 6;          // k / (KH*KW) with KH*KW == 64
    const int ky = (k & 63) >> 3;  // kx spans the 8-element run
    bf16x8 bfr[NF];
#pragma unroll
    for (int j = 0; j < NF; ++j) {
      bfr[j] = *reinterpret_cast<const bf16x8*>(&W[(j * 16 + ln) * K + k]);
    }
    const int rowoff = (c * LROWS + ky) * IWP;
#pragma unroll
    for (int f = 0; f < MAX_MF; ++f) {
      if (f >= nmf) break;
      const __bf16* p = &img[rowoff + aoff[f]];
      bf16x8 a;
      const bf16x4 alo = *reinterpret_cast<const bf16x4*>(p);
      const bf16x4 ahi = *reinterpret_cast<const bf16x4*>(p + 4);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        a[i] = alo[i];
        a[4 + i] = ahi[i];
      }
#pragma unroll
      for (int j = 0; j < NF; ++j) {
        acc[f][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr[j],
                                                            acc[f][j], 0, 0, 0);
      }
    }
  }

  // Epilogue: +bias, ReLU, NHWC bf16.
  __bf16* obase = out + ((int64_t)s * OH + oy0) * OW * CO;
  int fi = 0;
  for (int f = wave; f < MF; f += kWaves, ++fi) {
#pragma unroll
    for (int j = 0; j < NF; ++j) {
      const int cch = j * 16 + ln;
      const float b = bias[cch];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = f * 16 + lg * 4 + r;
        if (m >= mvalid) continue;
        const float v = acc[fi][j][r] + b;
        obase[(int64_t)m * CO + cch] = (__bf16)(v > 0.f ? v : 0.f);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Generic NHWC bf16 implicit-GEMM conv (fwd and dgrad), k-order (ky,kx,c).
//
// MODE 0: fwd, NHWC bf16 out, +bias +ReLU.
// MODE 1: fwd, fp32 NCHW-flat out (final trunk layer feeding the fc), +b +ReLU.
// MODE 2: dgrad: no bias, epilogue multiplies by relu'(act) (act = the saved
//         post-ReLU NHWC activation of the layer whose input grad this is).
//
// PAD/DIL describe the LOGICAL input the kernel convolves (IHL x IWL): for
// dgrad the real dY tile is zero-padded by PAD and zero-dilated by DIL at
// staging time, turning stride-S backward into a stride-1 forward conv with
// rotated weights.
// ---------------------------------------------------------------------------

template <int CI, int IHR, int IWR, int PAD, int DIL, int KH, int KW, int ST,
          int CO, int OH, int OW, int SB, int OYT, int MODE, int NT = kThreads>
__global__ __launch_bounds__(NT) void conv_nhwc_kernel(
    const __bf16* __restrict__ in,   // [N, IHR, IWR, CI]
    const __bf16* __restrict__ W,    // [CO, KH*KW*CI] (ky,kx,c)-major
    const float* __restrict__ bias,  // [CO] (MODE 0/1)
    const __bf16* __restrict__ act,  // [N, OH, OW, CO] (MODE 2)
    void* __restrict__ out, int N) {
  // Logical tile sized from the OUTPUT window: when (input-kernel)%stride
  // != 0 the forward drops a trailing row/column, so the transposed-conv
  // (dgrad) windows reach one element past the dilated input — those
  // positions are zeros via the staging guards.
  constexpr int IWL0 = DIL * (IWR - 1) + 1 + 2 * PAD;
  constexpr int IWL1 = (OW - 1) * ST + KW;
  constexpr int IWL = IWL0 > IWL1 ? IWL0 : IWL1;
  constexpr int ROWE = IWL * CI;           // elements per logical row
  constexpr int LROWS = (OYT - 1) * ST + KH;  // staged logical rows
  constexpr int K = KH * KW * CI;
  constexpr int KP = ((K + 31) / 32) * 32;  // W is packed zero-padded to KP
  constexpr int M_BLK = SB * OYT * OW;
  constexpr int MF = (M_BLK + 15) / 16;
  constexpr int NF = CO / 16;
  constexpr int NW = NT / 64;
  constexpr int MAX_MF = (MF + NW - 1) / NW;
  constexpr int BANDS = (OH + OYT - 1) / OYT;
  static_assert(ROWE % 8 == 0, "LDS rows must be whole 16 B chunks");
  // A-fragments read 8 contiguous (kx*CI+c) elements: runs must not cross
  // a ky row. When KW*CI is not a multiple of 32, a 32-wide K-step spans
  // two ky rows and ky becomes per-lane-group instead of wave-uniform.
  static_assert((KW * CI) % 8 == 0, "K-runs must not cross ky rows");
  constexpr bool kKyUniform = ((KW * CI) % 32) == 0;
  static_assert(CI % 8 == 0 || (PAD == 0 && DIL == 1),
                "pad/dilate staging assumes chunks within one x");
  static_assert(SB == 1 || OYT == OH, "multi-sample blocks stage full rows");

  extern __shared__ char smem[];  // SB * LROWS * ROWE bf16, swizzled

  const int g = blockIdx.x;
  const int s0 = (g / BANDS) * SB;
  const int band = g % BANDS;
  const int oy0 = band * OYT;
  const int ly0 = oy0 * ST;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int ln = lane & 15;
  const int lg = lane >> 4;

  // ---- stage the logical input tile (zeros for pad/dilation gaps) ----
  {
    constexpr int NCHUNK = SB * LROWS * ROWE / 8;
    for (int idx = tid; idx < NCHUNK; idx += NT) {
      const int e0 = idx * 8;
      const int sl = e0 / (LROWS * ROWE);
      const int rem = e0 % (LROWS * ROWE);
      const int s = s0 + sl;
      bf16x8 v = {};
      if (s < N) {
        if (PAD == 0 && DIL == 1) {
          // Contiguous rows: LDS row r is real row ly0+r.
          v = *reinterpret_cast<const bf16x8*>(
              &in[(int64_t)(s * IHR + ly0) * ROWE + rem]);
        } else {
          const int ly = rem / ROWE + ly0;
          const int xo = rem % ROWE;
          const int x = xo / CI, c = xo % CI;
          const int ry = ly - PAD, rx = x - PAD;
          if (ry >= 0 && rx >= 0 && ry % DIL == 0 && rx % DIL == 0 &&
              ry / DIL < IHR && rx / DIL < IWR) {
            v = *reinterpret_cast<const bf16x8*>(
                &in[((int64_t)(s * IHR + ry / DIL) * IWR + rx / DIL) * CI + c]);
          }
        }
      }
      lds_write8_swz(smem, e0, v);
    }
  }
  __syncthreads();

  const int sv = min(SB, N - s0);
  const int ov = (SB > 1) ? OYT : min(OYT, OH - oy0);
  const int mvalid = (SB > 1) ? sv * OYT * OW : ov * OW;

  int aoff[MAX_MF];
  int nmf = 0;
  for (int f = wave; f < MF; f += NW, ++nmf) {
    const int m = min(f * 16 + ln, M_BLK - 1);
    const int sl = m / (OYT * OW);
    const int rm = m % (OYT * OW);
    const int oy = rm / OW, ox = rm % OW;
    aoff[nmf] = (sl * LROWS + oy * ST) * ROWE + ox * ST * CI;
  }

  f32x4 acc[MAX_MF][NF];
#pragma unroll
  for (int f = 0; f < MAX_MF; ++f)
#pragma unroll
    for (int j = 0; j < NF; ++j) acc[f][j] = {0.f, 0.f, 0.f, 0.f};

#pragma unroll
  for (int ks = 0; ks < KP / 32; ++ks) {
    const int k0 = ks * 32;
    bf16x8 bfr[NF];
#pragma unroll
    for (int j = 0; j < NF; ++j) {
      bfr[j] = *reinterpret_cast<const bf16x8*>(
          &W[(int64_t)(j * 16 + ln) * KP + k0 + 8 * lg]);
    }
    int rowoff;
    if (kKyUniform) {
      const int ky = k0 / (KW * CI);
      rowoff = ky * ROWE + k0 % (KW * CI) + 8 * lg;
    } else {
      // Per-lane-group ky; K-pad lanes read LDS offset 0 (finite) and the
      // zero-padded W columns annihilate the products.
      const int k = k0 + 8 * lg;
      rowoff = (KP == K || k < K) ? (k / (KW * CI)) * ROWE + k % (KW * CI) : 0;
    }
#pragma unroll
    for (int f = 0; f < MAX_MF; ++f) {
      if (f >= nmf) break;
      const bf16x8 a = lds_read8_swz(smem, aoff[f] + rowoff);
#pragma unroll
      for (int j = 0; j < NF; ++j) {
        acc[f][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr[j],
                                                            acc[f][j], 0, 0, 0);
      }
    }
  }

  // ---- epilogue ----
  int fi = 0;
  for (int f = wave; f < MF; f += NW, ++fi) {
#pragma unroll
    for (int j = 0; j < NF; ++j) {
      const int cch = j * 16 + ln;
      const float b = (MODE == 2 || MODE == 3) ? 0.f : bias[cch];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = f * 16 + lg * 4 + r;
        if (m >= mvalid) continue;
        const int sl = m / (OYT * OW);
        const int rm = m % (OYT * OW);
        const int oy = rm / OW + oy0, ox = rm % OW;
        const int64_t s = s0 + sl;
        float v = acc[fi][j][r] + b;
        if (MODE == 2) {
          const __bf16 a = act[((s * OH + oy) * OW + ox) * CO + cch];
          reinterpret_cast<__bf16*>(out)[((s * OH + oy) * OW + ox) * CO + cch] =
              (float)a > 0.f ? (__bf16)v : (__bf16)0.f;
        } else if (MODE == 1) {
          v = v > 0.f ? v : 0.f;
          reinterpret_cast<float*>(out)[s * (CO * OH * OW) + cch * (OH * OW) +
                                        oy * OW + ox] = v;
        } else {
          if (MODE == 0) v = v > 0.f ? v : 0.f;  // 3: plain dgrad, 5: +bias
          reinterpret_cast<__bf16*>(out)[((s * OH + oy) * OW + ox) * CO + cch] =
              (__bf16)v;
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// ReLU-mask + NCHW-flat -> NHWC transpose for the trunk output grad.
// ---------------------------------------------------------------------------

template <int CO, int OH, int OW>
__global__ void mask_d3_kernel(const float* __restrict__ dflat,
                               const float* __restrict__ outflat,
                               __bf16* __restrict__ d3m, int64_t total) {
  constexpr int OHW = OH * OW;
  for (int64_t e = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; e < total;
       e += (int64_t)gridDim.x * blockDim.x) {
    const int64_t s = e / (CO * OHW);
    const int rem = (int)(e % (CO * OHW));
    const int pos = rem / CO;  // oy*OW + ox
    const int c = rem % CO;
    const int64_t flat = s * (CO * OHW) + c * OHW + pos;
    const float v = outflat[flat] > 0.f ? dflat[flat] : 0.f;
    d3m[e] = (__bf16)v;
  }
}

// ---------------------------------------------------------------------------
// wgrad: dW[n][ky][kx*CI+c] = sum_m dY[m][n] * X[(m@ky)][kx*CI+c], reduced
// over M = N*OH*OW in MC-sized chunks (fp32 partials, no atomics).
// grid = (KH slices) x (M chunks); each block loops its chunk 32 rows at a
// time: transpose-stage [CO][32] of dY and [KW*CI][32] of X, one MFMA
// K-step (the reduction), accumulate output fragments in registers.
// X_U8: gather from u8 NCHW frames ((c,kx) k-order) instead of NHWC bf16.
// ---------------------------------------------------------------------------

template <int CI, int XH, int XW, int KH, int KW, int ST, int CO, int OH,
          int OW, int MC, bool X_U8, int PAD = 0,
          int KWCP = ((KW * CI + 15) / 16) * 16>
__global__ __launch_bounds__(kThreads) void wgrad_kernel(
    const void* __restrict__ xin,    // NHWC bf16 [N,XH,XW,CI] or u8 NCHW
    const __bf16* __restrict__ dy,   // [N, OH, OW, CO]
    float* __restrict__ partials,    // [nchunks, KH, CO, KWCP]
    float* __restrict__ db_partials,  // [nchunks, CO]
    int N) {
  constexpr int KWC = KW * CI;  // real K-columns; KWCP pads to MFMA tiles
  // 128 B staging rows + the conv kernel's XOR swizzle: the scalar
  // transpose writes (a column per lane) and the strided fragment reads
  // otherwise serialize on the 32 LDS banks (measured ~1 conflict per
  // VALU op at MPAD=40, profiles/trunk_mfma_pmc_r2.md).
  constexpr int MPAD = 64;
  constexpr int MSTEP = 64;  // two MFMA k-chunks per staging round
  constexpr int OF = (CO / 16) * (KWCP / 16);
  constexpr int PER_WAVE = (OF + kWaves - 1) / kWaves;
  static_assert(KWC % 8 == 0, "KW*CI must be whole 16 B chunks");
  static_assert(KWCP % 16 == 0 && KWCP >= KWC, "bad KWCP");
  static_assert(CO % 16 == 0, "CO must tile by 16");

  __shared__ char sDYT[CO * MPAD * 2];
  __shared__ char sXT[KWCP * MPAD * 2];
  auto stage = [](char* buf, int row, int col, __bf16 v) {
    *reinterpret_cast<__bf16*>(buf + swz((row * MPAD + col) * 2)) = v;
  };

  const int ky = blockIdx.x;
  const int chunk = blockIdx.y;
  const int64_t m0 = (int64_t)chunk * MC;
  const int64_t M = (int64_t)N * OH * OW;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int ln = lane & 15;
  const int lg = lane >> 4;

  f32x4 acc[PER_WAVE];
#pragma unroll
  for (int i = 0; i < PER_WAVE; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};
  float db = 0.f;
  // Zero the K-column pad rows once (staging never writes them).
  for (int i = tid; i < (KWCP - KWC) * MPAD; i += kThreads) {
    stage(sXT, KWC + i / MPAD, i % MPAD, (__bf16)0.f);
  }
  if (KWCP != KWC) __syncthreads();

  const int64_t mend = (m0 + MC < M) ? m0 + MC : M;
  for (int64_t ms = m0; ms < mend; ms += MSTEP) {
    // ---- transpose-stage dY[ms..ms+MSTEP) ----
    for (int idx = tid; idx < MSTEP * (CO / 8); idx += kThreads) {
      const int mm = idx / (CO / 8);
      const int ch = (idx % (CO / 8)) * 8;
      bf16x8 v = {};
      if (ms + mm < M) {
        v = *reinterpret_cast<const bf16x8*>(&dy[(ms + mm) * CO + ch]);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) stage(sDYT, ch + i, mm, v[i]);
    }
    // ---- transpose-stage X spans for this ky ----
    if (X_U8) {
      // u8 NCHW frames; k-order (c, kx), span per (m,c) = 8 px at stride 1.
      const uint8_t* xu = reinterpret_cast<const uint8_t*>(xin);
      for (int idx = tid; idx < MSTEP * CI; idx += kThreads) {
        const int mm = idx / CI;
        const int c = idx % CI;
        const int64_t m = ms + mm;
        float px[KW];
#pragma unroll
        for (int i = 0; i < KW; ++i) px[i] = 0.f;
        if (m < M) {
          const int64_t s = m / (OH * OW);
          const int rm = (int)(m % (OH * OW));
          const int oy = rm / OW, ox = rm % OW;
          const uint8_t* src =
              xu + ((s * CI + c) * XH + oy * ST + ky) * XW + ox * ST;
          // 4 B-aligned (ST==4, 84-px rows): two u32 loads.
          const uint32_t plo = *reinterpret_cast<const uint32_t*>(src);
          const uint32_t phi = *reinterpret_cast<const uint32_t*>(src + 4);
#pragma unroll
          for (int i = 0; i < 4; ++i) {
            px[i] = (1.0f / 255.0f) * ((plo >> (8 * i)) & 0xffu);
            px[4 + i] = (1.0f / 255.0f) * ((phi >> (8 * i)) & 0xffu);
          }
        }
#pragma unroll
        for (int i = 0; i < KW; ++i) stage(sXT, c * KW + i, mm, (__bf16)px[i]);
      }
    } else {
      const __bf16* xb = reinterpret_cast<const __bf16*>(xin);
      for (int idx = tid; idx < MSTEP * (KWC / 8); idx += kThreads) {
        const int mm = idx / (KWC / 8);
        const int ch = (idx % (KWC / 8)) * 8;
        const int64_t m = ms + mm;
        bf16x8 v = {};
        if (m < M) {
          const int64_t s = m / (OH * OW);
          const int rm = (int)(m % (OH * OW));
          const int oy = rm / OW, ox = rm % OW;
          const int iy = oy * ST + ky - PAD;
          if (PAD == 0) {
            v = *reinterpret_cast<const bf16x8*>(
                &xb[((s * XH + iy) * XW + ox * ST) * CI + ch]);
          } else if (iy >= 0 && iy < XH) {
            // Padded convs: the 8-element chunk may straddle the x border;
            // guard per element (staging only, not the hot K-loop).
#pragma unroll
            for (int i = 0; i < 8; ++i) {
              const int e = ch + i;           // kx*CI + c within the span
              const int ix = ox * ST + e / CI - PAD;
              if (ix >= 0 && ix < XW) {
                v[i] = xb[((s * XH + iy) * XW + ix) * CI + e % CI];
              }
            }
          }
        }
#pragma unroll
        for (int i = 0; i < 8; ++i) stage(sXT, ch + i, mm, v[i]);
      }
    }
    __syncthreads();

    // ---- one MFMA K-step per output fragment ----
#pragma unroll
    for (int i = 0; i < PER_WAVE; ++i) {
      const int of = wave * PER_WAVE + i;
      if (of >= OF) break;
      const int ni = of / (KWCP / 16);
      const int ki = of % (KWCP / 16);
#pragma unroll
      for (int half = 0; half < MSTEP / 32; ++half) {
        const bf16x8 a = lds_read8_swz(
            sDYT, (ni * 16 + ln) * MPAD + half * 32 + 8 * lg);
        const bf16x8 b = lds_read8_swz(
            sXT, (ki * 16 + ln) * MPAD + half * 32 + 8 * lg);
        acc[i] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[i], 0, 0, 0);
      }
    }
    // db: wave 0 (ky==0 blocks only; one lane per output channel).
    if (ky == 0 && wave == 0 && lane < CO) {
#pragma unroll
      for (int ch = 0; ch < MSTEP / 8; ++ch) {
        const bf16x8 v = lds_read8_swz(sDYT, lane * MPAD + ch * 8);
#pragma unroll
        for (int i = 0; i < 8; ++i) db += (float)v[i];
      }
    }
    __syncthreads();
  }

  // ---- write fp32 partials ----
  float* pbase =
      partials + ((int64_t)chunk * KH + ky) * CO * KWCP;
#pragma unroll
  for (int i = 0; i < PER_WAVE; ++i) {
    const int of = wave * PER_WAVE + i;
    if (of >= OF) break;
    const int ni = of / (KWCP / 16);
    const int ki = of % (KWCP / 16);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      pbase[(ni * 16 + lg * 4 + r) * KWCP + ki * 16 + ln] = acc[i][r];
    }
  }
  if (ky == 0 && wave == 0 && lane < CO) {
    db_partials[(int64_t)chunk * CO + lane] = db;
  }
}

// Sum partials over the chunk axis. out[i] = sum_c partials[c*stride + i].
__global__ void reduce_partials_kernel(const float* __restrict__ partials,
                                       float* __restrict__ out, int nchunks,
                                       int64_t stride) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < stride;
       i += (int64_t)gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int c = 0; c < nchunks; ++c) acc += partials[c * stride + i];
    out[i] = acc;
  }
}

// ---------------------------------------------------------------------------
// Geometry table (84x84x4 AtariNet; full-res instantiations can be added
// alongside). All static so the template instantiations stay explicit.
// ---------------------------------------------------------------------------

struct G1 {  // conv1 8x8 s4, 4->32: 84x84 -> 20x20
  static constexpr int CI = 4, IH = 84, IW = 84, KH = 8, KW = 8, ST = 4;
  static constexpr int CO = 32, OH = 20, OW = 20, OYT = 20;
  static constexpr int IWP = 88;
  static constexpr size_t LDS = (size_t)CI * IH * IWP * 2;
};
struct G2 {  // conv2 4x4 s2, 32->64: 20x20 -> 9x9
  static constexpr int CI = 32, IH = 20, IW = 20, KH = 4, KW = 4, ST = 2;
  static constexpr int CO = 64, OH = 9, OW = 9, SB = 3, OYT = 9;
  static constexpr size_t LDS = (size_t)SB * ((OYT - 1) * ST + KH) * IW * CI * 2;
};
struct G3 {  // conv3 3x3 s1, 64->64: 9x9 -> 7x7
  static constexpr int CI = 64, IH = 9, IW = 9, KH = 3, KW = 3, ST = 1;
  static constexpr int CO = 64, OH = 7, OW = 7, SB = 6, OYT = 7;
  static constexpr size_t LDS = (size_t)SB * ((OYT - 1) * ST + KH) * IW * CI * 2;
};
struct D3 {  // dgrad conv3: dY [7,7,64] pad 2 -> dX [9,9,64]
  static constexpr int CI = 64, IHR = 7, IWR = 7, PAD = 2, DIL = 1;
  static constexpr int KH = 3, KW = 3, ST = 1, CO = 64, OH = 9, OW = 9;
  static constexpr int SB = 4, OYT = 9;
  static constexpr int IWL = DIL * (IWR - 1) + 1 + 2 * PAD;
  static constexpr size_t LDS = (size_t)SB * ((OYT - 1) * ST + KH) * IWL * CI * 2;
};
struct D2 {  // dgrad conv2: dY [9,9,64] dil 2 pad 3 -> dX [20,20,32]
  static constexpr int CI = 64, IHR = 9, IWR = 9, PAD = 3, DIL = 2;
  static constexpr int KH = 4, KW = 4, ST = 1, CO = 32, OH = 20, OW = 20;
  static constexpr int SB = 1, OYT = 20;
  static constexpr int IWL = DIL * (IWR - 1) + 1 + 2 * PAD;
  static constexpr size_t LDS = (size_t)SB * ((OYT - 1) * ST + KH) * IWL * CI * 2;
};

// Full-resolution (210x160x3) geometry, BASELINE config 5. Output-row
// banding keeps every LDS tile under 2-blocks/CU budget.
struct F1 {  // conv1 8x8 s4, 3->32: 210x160 -> 51x39
  static constexpr int CI = 3, IH = 210, IW = 160, KH = 8, KW = 8, ST = 4;
  static constexpr int CO = 32, OH = 51, OW = 39, OYT = 13;
  static constexpr int IWP = 160;
  static constexpr size_t LDS =
      (size_t)CI * ((OYT - 1) * ST + KH) * IWP * 2;
};
struct F2 {  // conv2 4x4 s2, 32->64: 51x39 -> 24x18
  static constexpr int CI = 32, IH = 51, IW = 39, KH = 4, KW = 4, ST = 2;
  static constexpr int CO = 64, OH = 24, OW = 18, SB = 1, OYT = 6;
  static constexpr size_t LDS =
      (size_t)((OYT - 1) * ST + KH) * IW * CI * 2;
};
struct F3 {  // conv3 3x3 s1, 64->64: 24x18 -> 22x16 (flat 22528)
  static constexpr int CI = 64, IH = 24, IW = 18, KH = 3, KW = 3, ST = 1;
  static constexpr int CO = 64, OH = 22, OW = 16, SB = 1, OYT = 6;
  static constexpr size_t LDS =
      (size_t)((OYT - 1) * ST + KH) * IW * CI * 2;
};
struct FD3 {  // dgrad conv3 FR: dY [22,16,64] pad 2 -> dX [24,18,64]
  static constexpr int CI = 64, IHR = 22, IWR = 16, PAD = 2, DIL = 1;
  static constexpr int KH = 3, KW = 3, ST = 1, CO = 64, OH = 24, OW = 18;
  static constexpr int SB = 1, OYT = 12;
  static constexpr int IWL = DIL * (IWR - 1) + 1 + 2 * PAD;
  static constexpr size_t LDS =
      (size_t)((OYT - 1) * ST + KH) * IWL * CI * 2;
};
struct FD2 {  // dgrad conv2 FR: dY [24,18,64] dil 2 pad 3 -> dX [51,39,32]
  static constexpr int CI = 64, IHR = 24, IWR = 18, PAD = 3, DIL = 2;
  static constexpr int KH = 4, KW = 4, ST = 1, CO = 32, OH = 51, OW = 39;
  static constexpr int SB = 1, OYT = 8;
  // Output-driven logical width (42): one past the dilated input, see the
  // kernel's IWL derivation.
  static constexpr int IWL = (OW - 1) * ST + KW;
  static constexpr size_t LDS =
      (size_t)((OYT - 1) * ST + KH) * IWL * CI * 2;
};

bool is_geom(const torch::Tensor& frames, int C, int H, int W) {
  return frames.size(1) == C && frames.size(2) == H && frames.size(3) == W;
}

void check_bf16(const torch::Tensor& t, const char* what) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16 &&
                  t.is_contiguous(),
              what, ": contiguous bf16 GPU tensor required");
}

}  // namespace

template <typename C1, typename C2, typename C3>
std::vector<torch::Tensor> trunk_fwd_impl(torch::Tensor frames,
                                          torch::Tensor w1, torch::Tensor b1,
                                          torch::Tensor w2, torch::Tensor b2,
                                          torch::Tensor w3, torch::Tensor b3,
                                          bool want_stash) {
  check_bf16(w1, "w1");
  check_bf16(w2, "w2");
  check_bf16(w3, "w3");
  auto fr = frames.contiguous();
  const int N = fr.size(0);
  auto stream = at::cuda::getCurrentCUDAStream();
  auto bfopt = w1.options();

  auto a1 = torch::empty({N, C1::OH, C1::OW, C1::CO}, bfopt);
  auto a2 = torch::empty({N, C2::OH, C2::OW, C2::CO}, bfopt);
  auto out3 = torch::empty({N, (int64_t)C3::CO * C3::OH * C3::OW},
                           bfopt.dtype(torch::kFloat32));
  if (N == 0) return {out3, a1, a2};

  constexpr int kBands1 = (C1::OH + C1::OYT - 1) / C1::OYT;
  hipLaunchKernelGGL(
      (conv1_u8_kernel<C1::CI, C1::IH, C1::IW, C1::KH, C1::KW, C1::ST,
                       C1::CO, C1::OH, C1::OW, C1::OYT>),
      dim3(N * kBands1), dim3(kThreads), C1::LDS, stream,
      fr.data_ptr<uint8_t>(),
      reinterpret_cast<const __bf16*>(w1.data_ptr()), b1.data_ptr<float>(),
      reinterpret_cast<__bf16*>(a1.data_ptr()), N);

  // Small (inference-sized) batches need grid = N x bands for occupancy
  // (measured: conv3 at batch ~100 was the top GPU kernel purely from
  // sample-group underfill).
  if (N < 1024 && C2::SB > 1) {
    constexpr size_t kLds2 = (size_t)C2::IH * C2::IW * C2::CI * 2;
    hipLaunchKernelGGL(
        (conv_nhwc_kernel<C2::CI, C2::IH, C2::IW, 0, 1, C2::KH, C2::KW,
                          C2::ST, C2::CO, C2::OH, C2::OW, 1, C2::OH, 0, 384>),
        dim3(N), dim3(384), kLds2, stream,
        reinterpret_cast<const __bf16*>(a1.data_ptr()),
        reinterpret_cast<const __bf16*>(w2.data_ptr()), b2.data_ptr<float>(),
        nullptr, a2.data_ptr(), N);
    constexpr size_t kLds3 = (size_t)C3::IH * C3::IW * C3::CI * 2;
    hipLaunchKernelGGL(
        (conv_nhwc_kernel<C3::CI, C3::IH, C3::IW, 0, 1, C3::KH, C3::KW,
                          C3::ST, C3::CO, C3::OH, C3::OW, 1, C3::OH, 1, 256>),
        dim3(N), dim3(256), kLds3, stream,
        reinterpret_cast<const __bf16*>(a2.data_ptr()),
        reinterpret_cast<const __bf16*>(w3.data_ptr()), b3.data_ptr<float>(),
        nullptr, out3.data_ptr(), N);
    return want_stash ? std::vector<torch::Tensor>{out3, a1, a2}
                      : std::vector<torch::Tensor>{out3};
  }

  constexpr int kBands2 = (C2::OH + C2::OYT - 1) / C2::OYT;
  hipLaunchKernelGGL(
      (conv_nhwc_kernel<C2::CI, C2::IH, C2::IW, 0, 1, C2::KH, C2::KW, C2::ST,
                        C2::CO, C2::OH, C2::OW, C2::SB, C2::OYT, 0>),
      dim3(((N + C2::SB - 1) / C2::SB) * kBands2), dim3(kThreads), C2::LDS,
      stream, reinterpret_cast<const __bf16*>(a1.data_ptr()),
      reinterpret_cast<const __bf16*>(w2.data_ptr()), b2.data_ptr<float>(),
      nullptr, a2.data_ptr(), N);

  constexpr int kBands3 = (C3::OH + C3::OYT - 1) / C3::OYT;
  hipLaunchKernelGGL(
      (conv_nhwc_kernel<C3::CI, C3::IH, C3::IW, 0, 1, C3::KH, C3::KW, C3::ST,
                        C3::CO, C3::OH, C3::OW, C3::SB, C3::OYT, 1>),
      dim3(((N + C3::SB - 1) / C3::SB) * kBands3), dim3(kThreads), C3::LDS,
      stream, reinterpret_cast<const __bf16*>(a2.data_ptr()),
      reinterpret_cast<const __bf16*>(w3.data_ptr()), b3.data_ptr<float>(),
      nullptr, out3.data_ptr(), N);

  if (!want_stash) return {out3};
  return {out3, a1, a2};
}

std::vector<torch::Tensor> conv_trunk_fwd(torch::Tensor frames,
                                          torch::Tensor w1, torch::Tensor b1,
                                          torch::Tensor w2, torch::Tensor b2,
                                          torch::Tensor w3, torch::Tensor b3,
                                          bool want_stash) {
  TORCH_CHECK(frames.is_cuda() && frames.scalar_type() == torch::kUInt8 &&
                  frames.dim() == 4,
              "conv_trunk_fwd: u8 [N,C,H,W] GPU frames required");
  if (is_geom(frames, 4, 84, 84)) {
    return trunk_fwd_impl<G1, G2, G3>(frames, w1, b1, w2, b2, w3, b3,
                                      want_stash);
  }
  if (is_geom(frames, 3, 210, 160)) {
    return trunk_fwd_impl<F1, F2, F3>(frames, w1, b1, w2, b2, w3, b3,
                                      want_stash);
  }
  TORCH_CHECK(false, "conv_trunk_fwd: unsupported geometry ", frames.sizes());
}

template <typename C3>
torch::Tensor mask_d3_impl(torch::Tensor d_out3, torch::Tensor out3) {
  auto d = d_out3.contiguous();
  auto o = out3.contiguous();
  const int64_t N = d.size(0);
  auto d3m = torch::empty({N, C3::OH, C3::OW, C3::CO},
                          d.options().dtype(torch::kBFloat16));
  const int64_t total = N * C3::CO * C3::OH * C3::OW;
  const int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
  hipLaunchKernelGGL((mask_d3_kernel<C3::CO, C3::OH, C3::OW>), dim3(blocks),
                     dim3(256), 0, at::cuda::getCurrentCUDAStream(),
                     d.data_ptr<float>(), o.data_ptr<float>(),
                     reinterpret_cast<__bf16*>(d3m.data_ptr()), total);
  return d3m;
}

torch::Tensor conv_trunk_mask_d3(torch::Tensor d_out3, torch::Tensor out3) {
  TORCH_CHECK(d_out3.is_cuda() && d_out3.scalar_type() == torch::kFloat32,
              "mask_d3: fp32 grad required");
  const int64_t per = d_out3.size(1);
  if (per == G3::CO * G3::OH * G3::OW) return mask_d3_impl<G3>(d_out3, out3);
  if (per == F3::CO * F3::OH * F3::OW) return mask_d3_impl<F3>(d_out3, out3);
  TORCH_CHECK(false, "mask_d3: unsupported feature size ", per);
}

template <typename D>
torch::Tensor dgrad_impl(torch::Tensor dy, torch::Tensor wr,
                         torch::Tensor act) {
  const int N = dy.size(0);
  auto dx = torch::empty_like(act);
  constexpr int kBands = (D::OH + D::OYT - 1) / D::OYT;
  hipLaunchKernelGGL(
      (conv_nhwc_kernel<D::CI, D::IHR, D::IWR, D::PAD, D::DIL, D::KH, D::KW,
                        D::ST, D::CO, D::OH, D::OW, D::SB, D::OYT, 2>),
      dim3(((N + D::SB - 1) / D::SB) * kBands), dim3(kThreads), D::LDS,
      at::cuda::getCurrentCUDAStream(),
      reinterpret_cast<const __bf16*>(dy.data_ptr()),
      reinterpret_cast<const __bf16*>(wr.data_ptr()), nullptr,
      reinterpret_cast<const __bf16*>(act.data_ptr()), dx.data_ptr(), N);
  return dx;
}

torch::Tensor conv_trunk_dgrad3(torch::Tensor d3m, torch::Tensor w3r,
                                torch::Tensor a2) {
  check_bf16(d3m, "d3m");
  check_bf16(w3r, "w3r");
  check_bf16(a2, "a2");
  if (d3m.size(1) == D3::IHR && d3m.size(2) == D3::IWR) {
    return dgrad_impl<D3>(d3m, w3r, a2);
  }
  if (d3m.size(1) == FD3::IHR && d3m.size(2) == FD3::IWR) {
    return dgrad_impl<FD3>(d3m, w3r, a2);
  }
  TORCH_CHECK(false, "dgrad3: unsupported geometry ", d3m.sizes());
}

torch::Tensor conv_trunk_dgrad2(torch::Tensor d2, torch::Tensor w2r,
                                torch::Tensor a1) {
  check_bf16(d2, "d2");
  check_bf16(w2r, "w2r");
  check_bf16(a1, "a1");
  if (d2.size(1) == D2::IHR && d2.size(2) == D2::IWR) {
    return dgrad_impl<D2>(d2, w2r, a1);
  }
  if (d2.size(1) == FD2::IHR && d2.size(2) == FD2::IWR) {
    return dgrad_impl<FD2>(d2, w2r, a1);
  }
  TORCH_CHECK(false, "dgrad2: unsupported geometry ", d2.sizes());
}

namespace {

template <int CI, int XH, int XW, int KH, int KW, int ST, int CO, int OH,
          int OW, int MC, bool X_U8, int PAD = 0>
std::vector<torch::Tensor> run_wgrad(const torch::Tensor& x,
                                     const torch::Tensor& dy) {
  constexpr int KWC = ((KW * CI + 15) / 16) * 16;  // padded (kernel KWCP)
  const int N = dy.size(0);
  const int64_t M = (int64_t)N * OH * OW;
  const int nchunks = (int)((M + MC - 1) / MC);
  auto fopt = dy.options().dtype(torch::kFloat32);
  auto partials = torch::empty({(int64_t)nchunks, KH, CO, KWC}, fopt);
  auto db_partials = torch::empty({(int64_t)nchunks, CO}, fopt);
  // Partial layout per chunk is [KH][CO][KWC]; the reduced dW keeps that
  // shape and the Python wrapper permutes to PyTorch [CO,CI,KH,KW].
  auto dW = torch::empty({KH, CO, KWC}, fopt);
  auto db = torch::empty({CO}, fopt);
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(
      (wgrad_kernel<CI, XH, XW, KH, KW, ST, CO, OH, OW, MC, X_U8, PAD>),
      dim3(KH, nchunks), dim3(kThreads), 0, stream,
      x.data_ptr(), reinterpret_cast<const __bf16*>(dy.data_ptr()),
      partials.data_ptr<float>(), db_partials.data_ptr<float>(), N);
  // Reduce over chunks. The partial layout is [chunk][KH*CO*KWC]; out dW is
  // the per-chunk tail shape.
  const int64_t stride = (int64_t)KH * CO * KWC;
  hipLaunchKernelGGL(reduce_partials_kernel,
                     dim3((int)std::min<int64_t>((stride + 255) / 256, 1024)),
                     dim3(256), 0, stream, partials.data_ptr<float>(),
                     dW.data_ptr<float>(), nchunks, stride);
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(1), dim3(256), 0, stream,
                     db_partials.data_ptr<float>(), db.data_ptr<float>(),
                     nchunks, CO);
  return {dW, db};
}

}  // namespace

std::vector<torch::Tensor> conv_trunk_wgrad1(torch::Tensor frames,
                                             torch::Tensor d1) {
  TORCH_CHECK(frames.is_cuda() && frames.scalar_type() == torch::kUInt8,
              "wgrad1: u8 frames");
  check_bf16(d1, "d1");
  auto fr = frames.contiguous();
  if (is_geom(fr, 4, 84, 84)) {
    return run_wgrad<G1::CI, G1::IH, G1::IW, G1::KH, G1::KW, G1::ST, G1::CO,
                     G1::OH, G1::OW, 2048, true>(fr, d1);
  }
  if (is_geom(fr, 3, 210, 160)) {
    return run_wgrad<F1::CI, F1::IH, F1::IW, F1::KH, F1::KW, F1::ST, F1::CO,
                     F1::OH, F1::OW, 2048, true>(fr, d1);
  }
  TORCH_CHECK(false, "wgrad1: unsupported geometry ", fr.sizes());
}

std::vector<torch::Tensor> conv_trunk_wgrad2(torch::Tensor a1,
                                             torch::Tensor d2) {
  check_bf16(a1, "a1");
  check_bf16(d2, "d2");
  if (a1.size(1) == G2::IH && a1.size(2) == G2::IW) {
    return run_wgrad<G2::CI, G2::IH, G2::IW, G2::KH, G2::KW, G2::ST, G2::CO,
                     G2::OH, G2::OW, 2048, false>(a1, d2);
  }
  if (a1.size(1) == F2::IH && a1.size(2) == F2::IW) {
    return run_wgrad<F2::CI, F2::IH, F2::IW, F2::KH, F2::KW, F2::ST, F2::CO,
                     F2::OH, F2::OW, 2048, false>(a1, d2);
  }
  TORCH_CHECK(false, "wgrad2: unsupported geometry ", a1.sizes());
}

std::vector<torch::Tensor> conv_trunk_wgrad3(torch::Tensor a2,
                                             torch::Tensor d3m) {
  check_bf16(a2, "a2");
  check_bf16(d3m, "d3m");
  if (a2.size(1) == G3::IH && a2.size(2) == G3::IW) {
    return run_wgrad<G3::CI, G3::IH, G3::IW, G3::KH, G3::KW, G3::ST, G3::CO,
                     G3::OH, G3::OW, 1024, false>(a2, d3m);
  }
  if (a2.size(1) == F3::IH && a2.size(2) == F3::IW) {
    return run_wgrad<F3::CI, F3::IH, F3::IW, F3::KH, F3::KW, F3::ST, F3::CO,
                     F3::OH, F3::OW, 1024, false>(a2, d3m);
  }
  TORCH_CHECK(false, "wgrad3: unsupported geometry ", a2.sizes());
}

// ---------------------------------------------------------------------------
// Deep IMPALA ResNet 3x3 s1 p1 convs (bf16 channels_last activations).
// Geometry table covers the 84x84x4 deep net: the first conv at 84x84
// (obs channels zero-padded to 8 so K-runs fill an A-fragment), then
// sections at 42x42 (16ch), 21x21 and 11x11 (32ch).
// ---------------------------------------------------------------------------

namespace {

template <int CI_, int HW, int CO_, int OYT_>
struct RGeom {
  static constexpr int CI = CI_, CO = CO_;
  static constexpr int IHR = HW, IWR = HW, PAD = 1, DIL = 1;
  static constexpr int KH = 3, KW = 3, ST = 1;
  static constexpr int OH = HW, OW = HW, SB = 1, OYT = OYT_;
  static constexpr int IWL = HW + 2;
  static constexpr size_t LDS = (size_t)((OYT - 1) * ST + KH) * IWL * CI * 2;
};

using RC_8_84_16 = RGeom<8, 84, 16, 7>;     // first conv, obs channels
                                            // zero-padded 4->8 (no dgrad:
                                            // frames carry no grad)
using RC_16_42_16 = RGeom<16, 42, 16, 7>;   // section-1 residual convs
using RC_16_42_32 = RGeom<16, 42, 32, 7>;   // section-2 feature conv
using RC_32_42_16 = RGeom<32, 42, 16, 7>;   // ... and its dgrad
using RC_32_21_32 = RGeom<32, 21, 32, 7>;   // section-2 res / section-3 feat
using RC_32_11_32 = RGeom<32, 11, 32, 11>;  // section-3 residual convs

template <typename G, int MODE>
torch::Tensor launch_rconv(const torch::Tensor& in, const torch::Tensor& w,
                           const float* bias_p, int64_t N) {
  auto out = torch::empty({N, G::OH, G::OW, G::CO}, w.options());
  constexpr int kBands = (G::OH + G::OYT - 1) / G::OYT;
  hipLaunchKernelGGL(
      (conv_nhwc_kernel<G::CI, G::IHR, G::IWR, G::PAD, G::DIL, G::KH, G::KW,
                        G::ST, G::CO, G::OH, G::OW, G::SB, G::OYT, MODE>),
      dim3((uint32_t)(N * kBands)), dim3(kThreads), G::LDS,
      at::cuda::getCurrentCUDAStream(),
      reinterpret_cast<const __bf16*>(in.data_ptr()),
      reinterpret_cast<const __bf16*>(w.data_ptr()), bias_p, nullptr,
      out.data_ptr(), (int)N);
  return out;
}

}  // namespace

// x: bf16 channels_last [N, CI, HW, HW] (NHWC storage); w: bf16
// [CO, KP] packed (ky,kx,c)-major zero-padded to KP = ceil32(9*CI);
// bias fp32 [CO] for fwd, empty for dgrad (which takes pre-rotated w).
torch::Tensor resnet_conv(torch::Tensor x, torch::Tensor w,
                          torch::Tensor bias, int64_t ci, int64_t hw,
                          int64_t co, bool fwd) {
  check_bf16(w, "resnet w");
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 &&
                  (x.dim() != 4 ||
                   x.is_contiguous(at::MemoryFormat::ChannelsLast)),
              "resnet_conv: channels_last bf16 GPU input required");
  const int64_t N = x.numel() / (ci * hw * hw);
  const float* bias_p =
      bias.numel() > 0 ? bias.data_ptr<float>() : nullptr;
  TORCH_CHECK(fwd == (bias_p != nullptr), "fwd needs bias, dgrad must not");
  if (N == 0) return torch::empty({0, hw, hw, co}, w.options());
  if (ci == 8 && hw == 84 && co == 16) {
    TORCH_CHECK(fwd, "first conv has no dgrad (frames carry no grad)");
    return launch_rconv<RC_8_84_16, 5>(x, w, bias_p, N);
  }
  if (ci == 16 && hw == 42 && co == 16) {
    return fwd ? launch_rconv<RC_16_42_16, 5>(x, w, bias_p, N)
               : launch_rconv<RC_16_42_16, 3>(x, w, bias_p, N);
  }
  if (ci == 16 && hw == 42 && co == 32) {
    TORCH_CHECK(fwd, "16->32 @42 is the feature conv (fwd only)");
    return launch_rconv<RC_16_42_32, 5>(x, w, bias_p, N);
  }
  if (ci == 32 && hw == 42 && co == 16) {
    TORCH_CHECK(!fwd, "32->16 @42 is the feature conv's dgrad");
    return launch_rconv<RC_32_42_16, 3>(x, w, bias_p, N);
  }
  if (ci == 32 && hw == 21 && co == 32) {
    return fwd ? launch_rconv<RC_32_21_32, 5>(x, w, bias_p, N)
               : launch_rconv<RC_32_21_32, 3>(x, w, bias_p, N);
  }
  if (ci == 32 && hw == 11 && co == 32) {
    return fwd ? launch_rconv<RC_32_11_32, 5>(x, w, bias_p, N)
               : launch_rconv<RC_32_11_32, 3>(x, w, bias_p, N);
  }
  TORCH_CHECK(false, "resnet_conv: unsupported geometry ci=", ci, " hw=", hw,
              " co=", co);
}

bool resnet_conv_supported(int64_t ci, int64_t hw, int64_t co) {
  return (ci == 8 && hw == 84 && co == 16) ||
         (ci == 16 && hw == 42 && (co == 16 || co == 32)) ||
         (ci == 32 && hw == 42 && co == 16) ||
         (ci == 32 && hw == 21 && co == 32) ||
         (ci == 32 && hw == 11 && co == 32);
}

// wgrad for the 3x3 s1 p1 convs; x/dy bf16 NHWC storage. Returns
// {dW [3, CO, KWCP] fp32, db [CO] fp32}.
std::vector<torch::Tensor> resnet_conv_wgrad(torch::Tensor x,
                                             torch::Tensor dy, int64_t ci,
                                             int64_t hw, int64_t co) {
  auto check_cl = [](const torch::Tensor& t, const char* what) {
    TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16 &&
                    (t.dim() != 4 ||
                     t.is_contiguous(at::MemoryFormat::ChannelsLast)),
                what, ": channels_last bf16 GPU tensor required");
  };
  check_cl(x, "resnet wgrad x");
  check_cl(dy, "resnet wgrad dy");
  if (ci == 8 && hw == 84 && co == 16) {
    return run_wgrad<8, 84, 84, 3, 3, 1, 16, 84, 84, 4096, false, 1>(x, dy);
  }
  if (ci == 16 && hw == 42 && co == 16) {
    return run_wgrad<16, 42, 42, 3, 3, 1, 16, 42, 42, 2048, false, 1>(x, dy);
  }
  if (ci == 16 && hw == 42 && co == 32) {
    return run_wgrad<16, 42, 42, 3, 3, 1, 32, 42, 42, 2048, false, 1>(x, dy);
  }
  if (ci == 32 && hw == 21 && co == 32) {
    return run_wgrad<32, 21, 21, 3, 3, 1, 32, 21, 21, 2048, false, 1>(x, dy);
  }
  if (ci == 32 && hw == 11 && co == 32) {
    return run_wgrad<32, 11, 11, 3, 3, 1, 32, 11, 11, 1024, false, 1>(x, dy);
  }
  TORCH_CHECK(false, "resnet_conv_wgrad: unsupported geometry");
}

}  // namespace tbamd