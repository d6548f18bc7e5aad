// Declarations for runtime_kernels.hip (gather_obs, fused_heads_sample).
#pragma once

#include <torch/extension.h>

#include <vector>

namespace tbruntime {

std::vector<torch::Tensor> gather_obs(torch::Tensor slab_frames,
                                      torch::Tensor slab_rew,
                                      torch::Tensor slab_done,
                                      torch::Tensor ids_cpu, int64_t bp,
                                      std::vector<int64_t> frame_shape);

std::vector<torch::Tensor> fused_heads_sample(torch::Tensor x,
                                              torch::Tensor rew,
                                              torch::Tensor policy_w,
                                              torch::Tensor policy_b,
                                              torch::Tensor base_w,
                                              torch::Tensor base_b, int64_t b,
                                              bool greedy, int64_t seed);

}  // namespace tbruntime
