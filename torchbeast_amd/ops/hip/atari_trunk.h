// Host API for the fused AtariNet conv-trunk kernels (atari_trunk.hip).
// Compiled into both _tbops (learner autograd op) and _tbruntime (C++
// inference engine).

#pragma once

#include <torch/extension.h>

#include <vector>

namespace tbamd {

// True if the fused kernel supports this frame geometry (intermediates must
// fit in the 160 KiB LDS of a gfx950 CU). 84x84x4 fits; 210x160x3 does not
// and falls back to library convs.
bool atari_trunk_supported(int64_t C, int64_t H, int64_t W);

// frames: [N, C, H, W] uint8 on GPU. Weights: PyTorch Conv2d layouts
// (w1 [32,C,8,8] s4, w2 [64,32,4,4] s2, w3 [64,64,3,3] s1), fp32.
// Returns {out3_flat [N, 64*H3*W3]} and, if save_for_backward,
// {out3_flat, out1 [N,32,H1,W1], out2 [N,64,H2,W2]}.
std::vector<torch::Tensor> atari_trunk_fwd(
    torch::Tensor frames, torch::Tensor w1, torch::Tensor b1,
    torch::Tensor w2, torch::Tensor b2, torch::Tensor w3, torch::Tensor b3,
    bool save_for_backward);

// d_out3: [N, 64*H3*W3] upstream gradient (pre relu-mask).
// Returns {dw1, db1, dw2, db2, dw3, db3}.

// bf16 MFMA GEMM (mfma_gemm.hip): C[M,N] f32 = A[M,K] @ B[N,K]^T.
torch::Tensor mfma_gemm(torch::Tensor A, torch::Tensor B);
torch::Tensor mfma_gemm_probe(torch::Tensor A, torch::Tensor B,
                              int64_t variant);
torch::Tensor mfma_gemm_v2(torch::Tensor A, torch::Tensor B);

}  // namespace tbamd
