// MFMA implicit-GEMM AtariNet conv trunk (gfx950). See conv_mfma.hip.
#pragma once

#include <torch/extension.h>

#include <vector>

namespace tbamd {

// Forward: u8 NCHW frames -> three conv layers -> flat fp32 features
// (NCHW flatten order, matching monobeast.py:552-562's view()).
// Returns {out3_flat_f32 [N,3136], a1 [N,20,20,32] bf16, a2 [N,9,9,64] bf16}
// (a1/a2 are the post-ReLU NHWC activations, stashed for backward).
std::vector<torch::Tensor> conv_trunk_fwd(torch::Tensor frames,
                                          torch::Tensor w1, torch::Tensor b1,
                                          torch::Tensor w2, torch::Tensor b2,
                                          torch::Tensor w3, torch::Tensor b3,
                                          bool want_stash);

// d_out3 fp32 [N,3136] * relu'(out3) -> NHWC bf16 [N,7,7,64].
torch::Tensor conv_trunk_mask_d3(torch::Tensor d_out3, torch::Tensor out3);

// dgrad conv3: d3m [N,7,7,64] bf16, w3r = rotated/permuted weight
// [64, 3*3*64] bf16, a2 = saved activation (relu mask). -> d2 [N,9,9,64].
torch::Tensor conv_trunk_dgrad3(torch::Tensor d3m, torch::Tensor w3r,
                                torch::Tensor a2);

// dgrad conv2 (stride 2, via dilated staging): d2 [N,9,9,64], w2r
// [32, 4*4*64] bf16, a1 mask. -> d1 [N,20,20,32] bf16.
torch::Tensor conv_trunk_dgrad2(torch::Tensor d2, torch::Tensor w2r,
                                torch::Tensor a1);

// wgrads. Layout of the returned dW matches the bf16 operand layout the
// forward consumes ((ky,kx,c) k-order for conv2/3, (c,ky,kx) for conv1);
// the Python wrapper permutes back to PyTorch [co,ci,kh,kw].
// Each returns {dW fp32, db fp32}.
std::vector<torch::Tensor> conv_trunk_wgrad1(torch::Tensor frames,
                                             torch::Tensor d1);
std::vector<torch::Tensor> conv_trunk_wgrad2(torch::Tensor a1,
                                             torch::Tensor d2);
std::vector<torch::Tensor> conv_trunk_wgrad3(torch::Tensor a2,
                                             torch::Tensor d3m);

// Deep IMPALA ResNet 3x3 s1 p1 convs over bf16 channels_last activations
// (reference models/resnet_monobeast.py feat_extract; see models/resnet.py).
// x: bf16 NHWC storage [N,CI,HW,HW]; w: bf16 [CO, ceil32(9*CI)] packed
// (ky,kx,c)-major, zero-padded; bias fp32 [CO] for fwd (empty for dgrad,
// which takes pre-rotated weights). Returns bf16 NHWC [N,CO,HW,HW].
torch::Tensor resnet_conv(torch::Tensor x, torch::Tensor w,
                          torch::Tensor bias, int64_t ci, int64_t hw,
                          int64_t co, bool fwd);
bool resnet_conv_supported(int64_t ci, int64_t hw, int64_t co);
// {dW [3, CO, KWCP] fp32, db [CO] fp32}; KWCP = ceil16(3*CI).
std::vector<torch::Tensor> resnet_conv_wgrad(torch::Tensor x,
                                             torch::Tensor dy, int64_t ci,
                                             int64_t hw, int64_t co);

}  // namespace tbamd
