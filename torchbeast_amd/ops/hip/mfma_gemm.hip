// Hand-written bf16 MFMA GEMM for gfx950 (CDNA4 matrix cores).
//
// C[M,N] (fp32) = A[M,K] (bf16, row-major) @ B[N,K]^T (bf16, row-major —
// the nn.Linear weight layout, so both operands stream K-contiguous rows).
//
// Structure (correctness-first; see profiles/mfma_gemm_bench.md for
// measured numbers): 64x64 tile per 256-thread workgroup (2x2 waves, one
// 32x32 sub-tile per wave = 2x2 fragments of v_mfma_f32_16x16x32_bf16),
// K-loop staging 64x32 A/B tiles in LDS with short8 (16 B/lane) vector
// loads. The C/D fragment layout is the HW-verified mapping
// col = lane & 15, row = (lane >> 4) * 4 + reg.
//
// The A/B input fragment layout on gfx950 is probed empirically (the two
// plausible mappings for the doubled-K instruction): variant 0 = lane
// holds 8 contiguous K (k = 8*(lane>>4) + i), variant 1 = two K=16 halves
// each mapped like the K=16 instruction (k = 4*(lane>>4) + i, +16 for the
// upper half). `mfma_gemm_probe` exposes both so a single GPU test run
// identifies the real one; `mfma_gemm` uses the verified variant.

#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <vector>

namespace tbamd {

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int kTileM = 64;
constexpr int kTileN = 64;
constexpr int kTileK = 32;

// lane -> 8 K-indices of the A/B fragment, for the probed layout variants.
template <int LAYOUT>
__device__ inline int frag_k(int lane, int i) {
  if (LAYOUT == 0) {
    return 8 * (lane >> 4) + i;  // contiguous 8
  }
  // Two K=16 halves, each with the classic 4-per-lane-group mapping.
  const int half = i / 4;
  return half * 16 + 4 * (lane >> 4) + (i & 3);
}

template <int LAYOUT, bool SWAP>
__global__ __launch_bounds__(256) void mfma_gemm_kernel(
    const __bf16* __restrict__ A,  // [M, K]
    const __bf16* __restrict__ B,  // [N, K]
    float* __restrict__ C,         // [M, N]
    int M, int N, int K) {
  __shared__ __bf16 sA[kTileM][kTileK];
  __shared__ __bf16 sB[kTileN][kTileK];

  const int tile_m = blockIdx.x * kTileM;
  const int tile_n = blockIdx.y * kTileN;
  const int tid = threadIdx.x;
  const int wave = tid / 64;   // 0..3 -> (wm, wn) in 2x2
  const int lane = tid % 64;
  const int wm = (wave >> 1) * 32;  // wave's 32x32 sub-tile origin
  const int wn = (wave & 1) * 32;

  f32x4 acc[2][2];
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int b = 0; b < 2; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < K; k0 += kTileK) {
    // Stage A/B tiles: 64 rows x 32 K of bf16 = 4 KiB each; 256 threads x
    // 16 B = one pass per tile.
    {
      const int row = tid / 4;          // 0..63
      const int kofs = (tid % 4) * 8;   // 0,8,16,24
      *reinterpret_cast<bf16x8*>(&sA[row][kofs]) =
          *reinterpret_cast<const bf16x8*>(
              &A[(int64_t)(tile_m + row) * K + k0 + kofs]);
      *reinterpret_cast<bf16x8*>(&sB[row][kofs]) =
          *reinterpret_cast<const bf16x8*>(
              &B[(int64_t)(tile_n + row) * K + k0 + kofs]);
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < kTileK; kk += 32) {
      // Fragments for this wave's 2x2 16x16 outputs.
      bf16x8 af[2], bf[2];
#pragma unroll
      for (int m = 0; m < 2; ++m) {
        const int row = wm + m * 16 + (lane & 15);
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          af[m][i] = sA[row][kk + frag_k<LAYOUT>(lane, i)];
        }
      }
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        const int col = wn + n * 16 + (lane & 15);
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          bf[n][i] = sB[col][kk + frag_k<LAYOUT>(lane, i)];
        }
      }
#pragma unroll
      for (int m = 0; m < 2; ++m) {
#pragma unroll
        for (int n = 0; n < 2; ++n) {
          if (SWAP) {
            acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                bf[n], af[m], acc[m][n], 0, 0, 0);
          } else {
            acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[m], bf[n], acc[m][n], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  // C/D layout (HW-verified): col = lane&15, row = (lane>>4)*4 + reg.
#pragma unroll
  for (int m = 0; m < 2; ++m) {
#pragma unroll
    for (int n = 0; n < 2; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = tile_m + wm + m * 16 + (lane >> 4) * 4 + r;
        const int col = tile_n + wn + n * 16 + (lane & 15);
        C[(int64_t)row * N + col] = acc[m][n][r];
      }
    }
  }
}

template <int LAYOUT, bool SWAP>
torch::Tensor launch(torch::Tensor A, torch::Tensor B) {
  const int M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16 &&
                  B.scalar_type() == torch::kBFloat16,
              "mfma_gemm: bf16 GPU tensors required");
  TORCH_CHECK(B.size(1) == K, "mfma_gemm: K mismatch");
  TORCH_CHECK(M % kTileM == 0 && N % kTileN == 0 && K % kTileK == 0,
              "mfma_gemm: M,N multiple of 64 and K multiple of 32 required");
  auto C = torch::empty({M, N}, A.options().dtype(torch::kFloat32));
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  hipLaunchKernelGGL((mfma_gemm_kernel<LAYOUT, SWAP>),
                     dim3(M / kTileM, N / kTileN), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     reinterpret_cast<const __bf16*>(Ac.data_ptr()),
                     reinterpret_cast<const __bf16*>(Bc.data_ptr()),
                     C.data_ptr<float>(), M, N, K);
  return C;
}

// ---------------------------------------------------------------------------
// v2: 128x128 tile, 4 waves (64x64 sub-tile each, 4x4 fragments), direct
// global->LDS staging via __builtin_amdgcn_global_load_lds (width 16). The
// LDS destination of global_load_lds is wave-uniform-base + lane*16, so the
// [128][32] bf16 tiles stay linear row-major and each wave stages a 32-row
// band (lane -> row = lane/4, 16B chunk = lane%4).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void mfma_gemm_v2_kernel(
    const __bf16* __restrict__ A,  // [M, K]
    const __bf16* __restrict__ B,  // [N, K]
    float* __restrict__ C,         // [M, N]
    int M, int N, int K) {
  constexpr int BM = 128, BN = 128, BK = 32;
  __shared__ __bf16 sA[BM][BK];
  __shared__ __bf16 sB[BN][BK];

  // (XCD-aware block swizzle was measured here and reverted: it pays only
  // when operands exceed the 256 MiB L3 — +10% at 8k per the CDNA4 guide —
  // and costs ~4% at the L3-resident sizes this kernel serves.)
  const int tile_m = blockIdx.x * BM;
  const int tile_n = blockIdx.y * BN;
  const int tid = threadIdx.x;
  const int wave = tid / 64;
  const int lane = tid % 64;
  const int wm = (wave >> 1) * 64;  // wave's 64x64 sub-tile origin
  const int wn = (wave & 1) * 64;

  f32x4 acc[4][4];
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 4; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  // Staging geometry: each wave loads a 32-row band of each tile; a band is
  // 32 rows x 64 B = 2 KiB = two 1 KiB (64 lanes x 16 B) issues.
  const int band = wave * 32;
  const int lrow = lane / 4;          // 0..15
  const int lchunk = (lane % 4) * 8;  // bf16 elements: 0, 8, 16, 24


  for (int k0 = 0; k0 < K; k0 += BK) {
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      // Hardware semantics: lane l's 16 B land at lds_base + l*16 (the
      // base is wave-uniform). With row = lane/4 and chunk = lane%4 the
      // linear row-major tile offset equals lane*16 exactly, so the base
      // is the start of this 16-row half-band.
      const int row = band + half * 16 + lrow;
      typedef const __attribute__((address_space(1))) uint32_t* gptr_t;
      typedef __attribute__((address_space(3))) uint32_t* lptr_t;
      __builtin_amdgcn_global_load_lds(
          (gptr_t)(&A[(int64_t)(tile_m + row) * K + k0 + lchunk]),
          (lptr_t)(&sA[band + half * 16][0]), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (gptr_t)(&B[(int64_t)(tile_n + row) * K + k0 + lchunk]),
          (lptr_t)(&sB[band + half * 16][0]), 16, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

    // 16 MFMA per wave per K-step: 4x4 fragments of 16x16x32.
    bf16x8 af[4], bf[4];
#pragma unroll
    for (int m = 0; m < 4; ++m) {
      af[m] = *reinterpret_cast<const bf16x8*>(
          &sA[wm + m * 16 + (lane & 15)][8 * (lane >> 4)]);
    }
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      bf[n] = *reinterpret_cast<const bf16x8*>(
          &sB[wn + n * 16 + (lane & 15)][8 * (lane >> 4)]);
    }
#pragma unroll
    for (int m = 0; m < 4; ++m) {
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[m], bf[n], acc[m][n], 0, 0, 0);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = tile_m + wm + m * 16 + (lane >> 4) * 4 + r;
        const int col = tile_n + wn + n * 16 + (lane & 15);
        C[(int64_t)row * N + col] = acc[m][n][r];
      }
    }
  }
}

}  // namespace

torch::Tensor mfma_gemm_v2(torch::Tensor A, torch::Tensor B) {
  const int M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == torch::kBFloat16 &&
                  B.scalar_type() == torch::kBFloat16,
              "mfma_gemm_v2: bf16 GPU tensors required");
  TORCH_CHECK(B.size(1) == K, "mfma_gemm_v2: K mismatch");
  TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 32 == 0,
              "mfma_gemm_v2: M,N multiple of 128, K multiple of 32");
  auto C = torch::empty({M, N}, A.options().dtype(torch::kFloat32));
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  hipLaunchKernelGGL(mfma_gemm_v2_kernel, dim3(M / 128, N / 128), dim3(256),
                     0, at::cuda::getCurrentCUDAStream(),
                     reinterpret_cast<const __bf16*>(Ac.data_ptr()),
                     reinterpret_cast<const __bf16*>(Bc.data_ptr()),
                     C.data_ptr<float>(), M, N, K);
  return C;
}

// variant: 0 = (layout0, noswap), 1 = (layout1, noswap),
//          2 = (layout0, swap),   3 = (layout1, swap).
torch::Tensor mfma_gemm_probe(torch::Tensor A, torch::Tensor B,
                              int64_t variant) {
  switch (variant) {
    case 0: return launch<0, false>(A, B);
    case 1: return launch<1, false>(A, B);
    case 2: return launch<0, true>(A, B);
    case 3: return launch<1, true>(A, B);
    default: TORCH_CHECK(false, "unknown variant");
  }
}

// Dispatch: the 128-tile global_load_lds kernel when shapes allow (685 TF
// bf16 @4096^3 measured), else the 64-tile fallback (387 TF). Fragment
// K-layout note: any K-permutation shared by the A and B fragments cancels
// inside the dot product, so both probed layouts are exact; variant 0
// (contiguous-8) vectorizes the LDS reads and is the fast one.
torch::Tensor mfma_gemm(torch::Tensor A, torch::Tensor B) {
  if (A.size(0) % 128 == 0 && B.size(0) % 128 == 0) {
    return mfma_gemm_v2(A, B);
  }
  return mfma_gemm_probe(A, B, /*variant=*/0);
}

}  // namespace tbamd
