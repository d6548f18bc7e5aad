// GPU kernels for the C++ inference engine (gfx950).
//
// gather_obs: zero-copy assembly of an inference batch straight from the
// actors' pinned observation slab. Actors write frame/reward/done into
// their own slot (host, pinned); requests carry only the 4-byte slot id;
// this kernel reads the slab over the host link and materializes the
// batched GPU tensors in one launch. Replaces the per-request host-side
// nest cat + pageable H2D copy that dominated serve latency
// (profiles/PROFILE_r2.md: cat ~3.7 ms/batch at batch ~100).
//
// fused_heads_sample: fc-output [bp,512] + clipped reward -> policy
// logits, baseline and a Gumbel-argmax action sample in ONE kernel,
// writing results directly into pinned host output buffers (no separate
// D2H copies). Replaces ~12 ATen ops per serve.

#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <hiprand/hiprand_kernel.h>
#include <torch/extension.h>

#include <vector>

namespace tbruntime {

namespace {

__global__ void gather_obs_kernel(
    const uint8_t* __restrict__ slab_frames,  // host pinned [A, fsz]
    const float* __restrict__ slab_rew,       // host pinned [A]
    const uint8_t* __restrict__ slab_done,    // host pinned [A]
    const int* __restrict__ ids,              // device [b]
    int b, int bp, int fsz,
    uint8_t* __restrict__ out_frames,  // [bp, fsz]
    float* __restrict__ out_rew,       // [bp, 1] clamped to [-1, 1]
    float* __restrict__ out_nd) {      // [bp] 1 - done
  const int s = blockIdx.x;
  if (s >= bp) return;
  uint8_t* dst = out_frames + (int64_t)s * fsz;
  if (s >= b) {
    for (int i = threadIdx.x * 16; i < fsz; i += blockDim.x * 16) {
      *reinterpret_cast<uint4*>(dst + i) = uint4{0, 0, 0, 0};
    }
    if (threadIdx.x == 0) {
      out_rew[s] = 0.f;
      out_nd[s] = 1.f;
    }
    return;
  }
  const uint8_t* src = slab_frames + (int64_t)ids[s] * fsz;
  for (int i = threadIdx.x * 16; i < fsz; i += blockDim.x * 16) {
    *reinterpret_cast<uint4*>(dst + i) = *reinterpret_cast<const uint4*>(src + i);
  }
  if (threadIdx.x == 0) {
    const float r = slab_rew[ids[s]];
    out_rew[s] = r < -1.f ? -1.f : (r > 1.f ? 1.f : r);
    out_nd[s] = slab_done[ids[s]] ? 0.f : 1.f;
  }
}

// One workgroup per sample; the core vector ([x, rew], length D+1) is
// staged in LDS, then each wave computes head dot products.
__global__ __launch_bounds__(256) void heads_sample_kernel(
    const float* __restrict__ x,         // [bp, D] fc output (post-relu)
    const float* __restrict__ rew,       // [bp, 1]
    const float* __restrict__ policy_w,  // [A, D+1]
    const float* __restrict__ policy_b,  // [A]
    const float* __restrict__ base_w,    // [1, D+1]
    const float* __restrict__ base_b,    // [1]
    int b, int D, int A, uint64_t seed, int greedy,
    int64_t* __restrict__ out_action,  // host pinned [b]
    float* __restrict__ out_logits,    // host pinned [b, A]
    float* __restrict__ out_base) {    // host pinned [b]
  const int s = blockIdx.x;
  if (s >= b) return;
  extern __shared__ float s_core[];  // [D+1]
  const int Dc = D + 1;
  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    s_core[i] = x[(int64_t)s * D + i];
  }
  if (threadIdx.x == 0) s_core[D] = rew[s];
  __syncthreads();

  // One lane group per output row (A logits + 1 baseline): wave-strided
  // dot products, then lane-0 reduction via shuffles.
  __shared__ float s_logits[64];  // A <= 64
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int nwaves = blockDim.x >> 6;
  for (int row = wave; row <= A; row += nwaves) {
    const float* w = (row < A) ? policy_w + (int64_t)row * Dc : base_w;
    float acc = 0.f;
    for (int i = lane; i < Dc; i += 64) acc += s_core[i] * w[i];
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      acc += __shfl_down(acc, off, 64);
    }
    if (lane == 0) {
      if (row < A) {
        const float v = acc + policy_b[row];
        s_logits[row] = v;
        out_logits[(int64_t)s * A + row] = v;
      } else {
        out_base[s] = acc + base_b[0];
      }
    }
  }
  __syncthreads();

  if (threadIdx.x == 0) {
    int best = 0;
    if (greedy) {
      float bv = s_logits[0];
      for (int a = 1; a < A; ++a) {
        if (s_logits[a] > bv) {
          bv = s_logits[a];
          best = a;
        }
      }
    } else {
      hiprandStatePhilox4_32_10_t rng;
      hiprand_init(seed, s, 0, &rng);
      float bv = -1e30f;
      for (int a = 0; a < A; ++a) {
        float u = hiprand_uniform(&rng);
        u = u < 1e-20f ? 1e-20f : u;
        const float g = s_logits[a] - __logf(-__logf(u));
        if (g > bv) {
          bv = g;
          best = a;
        }
      }
    }
    out_action[s] = best;
  }
}

}  // namespace

// Host wrappers -------------------------------------------------------------

// Returns {frames [bp,C,H,W] u8 (GPU), rew [bp,1] f32 (GPU), nd [bp] f32}.
std::vector<torch::Tensor> gather_obs(torch::Tensor slab_frames,
                                      torch::Tensor slab_rew,
                                      torch::Tensor slab_done,
                                      torch::Tensor ids_cpu, int64_t bp,
                                      std::vector<int64_t> frame_shape) {
  const int b = ids_cpu.numel();
  auto dev = torch::Device(torch::kCUDA, at::cuda::current_device());
  auto opts = torch::TensorOptions().device(dev);
  // Pad the ids staging to the quantized batch size so the pinned-alloc
  // cache sees a handful of sizes (a fresh hipHostMalloc device-syncs).
  auto ids_pin = torch::zeros({bp}, torch::TensorOptions()
                                        .dtype(torch::kInt32)
                                        .pinned_memory(true));
  ids_pin.narrow(0, 0, b).copy_(ids_cpu.to(torch::kInt32).reshape({-1}));
  auto ids = ids_pin.to(dev, /*non_blocking=*/true);
  int64_t fsz = 1;
  for (auto d : frame_shape) fsz *= d;
  std::vector<int64_t> oshape = {bp};
  oshape.insert(oshape.end(), frame_shape.begin(), frame_shape.end());
  auto frames = torch::empty(oshape, opts.dtype(torch::kUInt8));
  auto rew = torch::empty({bp, 1}, opts.dtype(torch::kFloat32));
  auto nd = torch::empty({bp}, opts.dtype(torch::kFloat32));
  TORCH_CHECK(fsz % 16 == 0, "frame bytes must be divisible by 16");
  hipLaunchKernelGGL(gather_obs_kernel, dim3((uint32_t)bp), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     slab_frames.data_ptr<uint8_t>(),
                     slab_rew.data_ptr<float>(),
                     slab_done.data_ptr<uint8_t>(), ids.data_ptr<int>(),
                     b, (int)bp, (int)fsz, frames.data_ptr<uint8_t>(),
                     rew.data_ptr<float>(), nd.data_ptr<float>());
  return {frames, rew, nd};
}

// Writes action/logits/baseline into freshly allocated PINNED host tensors;
// caller syncs the stream before reading them.
std::vector<torch::Tensor> fused_heads_sample(torch::Tensor x,
                                              torch::Tensor rew,
                                              torch::Tensor policy_w,
                                              torch::Tensor policy_b,
                                              torch::Tensor base_w,
                                              torch::Tensor base_b, int64_t b,
                                              bool greedy, int64_t seed) {
  const int D = x.size(1);
  const int A = policy_w.size(0);
  TORCH_CHECK(A <= 64, "heads kernel supports up to 64 actions");
  TORCH_CHECK(policy_w.size(1) == D + 1 && base_w.size(1) == D + 1,
              "head weights must take [x, reward]");
  // Allocate pinned outputs at the quantized batch size (cache-friendly;
  // ragged sizes would hipHostMalloc + device-sync per serve), then hand
  // back narrowed views.
  const int64_t bq = (b + 63) / 64 * 64;
  auto hopts = torch::TensorOptions().pinned_memory(true);
  auto action = torch::empty({bq}, hopts.dtype(torch::kInt64)).narrow(0, 0, b);
  auto logits =
      torch::empty({bq, A}, hopts.dtype(torch::kFloat32)).narrow(0, 0, b);
  auto baseline =
      torch::empty({bq}, hopts.dtype(torch::kFloat32)).narrow(0, 0, b);
  const size_t lds = (size_t)(D + 1) * sizeof(float);
  hipLaunchKernelGGL(heads_sample_kernel, dim3((uint32_t)b), dim3(256), lds,
                     at::cuda::getCurrentCUDAStream(), x.data_ptr<float>(),
                     rew.data_ptr<float>(), policy_w.data_ptr<float>(),
                     policy_b.data_ptr<float>(), base_w.data_ptr<float>(),
                     base_b.data_ptr<float>(), (int)b, D, A,
                     (uint64_t)seed, greedy ? 1 : 0,
                     action.data_ptr<int64_t>(), logits.data_ptr<float>(),
                     baseline.data_ptr<float>());
  return {action, logits, baseline};
}

}  // namespace tbruntime
