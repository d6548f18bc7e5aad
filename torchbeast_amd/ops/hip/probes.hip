// Microbenchmark probes for design decisions (not on any hot path).
//
// barrier_probe: cost of a grid-wide barrier across NW workgroups, three
// implementations:
//   0: cooperative-groups grid.sync (the round-1 LSTM design's primitive)
//   1: sense-reversing barrier, __threadfence() for visibility
//   2: sense-reversing barrier, agent-scope atomics only
// Each kernel runs `iters` barriers; wall time / iters ~ barrier cost.
// A correctness check sums per-WG counters across the barrier so a broken
// coherence scheme shows up as a wrong checksum, not a fast lie.

#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_cooperative_groups.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

namespace cg = cooperative_groups;

namespace tbamd {

namespace {

#define DEVCHECK2(x) \
  TORCH_CHECK(x == hipSuccess, "HIP error: ", hipGetErrorString(x))

__device__ __forceinline__ void sense_barrier(int* count, int* sense,
                                              int nblocks, int* lsense,
                                              bool use_fence) {
  __syncthreads();
  if (threadIdx.x == 0) {
    if (use_fence) __threadfence();
    const int s = 1 - *lsense;
    *lsense = s;
    const int prev = __hip_atomic_fetch_add(count, 1, __ATOMIC_ACQ_REL,
                                            __HIP_MEMORY_SCOPE_AGENT);
    if (prev == nblocks - 1) {
      __hip_atomic_store(count, 0, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
      __hip_atomic_store(sense, s, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_AGENT);
    } else {
      while (__hip_atomic_load(sense, __ATOMIC_ACQUIRE,
                               __HIP_MEMORY_SCOPE_AGENT) != s) {
        __builtin_amdgcn_s_sleep(8);
      }
    }
    if (use_fence) __threadfence();
  }
  __syncthreads();
}

template <int VARIANT>
__global__ __launch_bounds__(256) void barrier_probe_kernel(
    int* __restrict__ count, int* __restrict__ sense, int* __restrict__ data,
    int iters, int nblocks, long long* __restrict__ checksum) {
  __shared__ int lsense;
  if (threadIdx.x == 0) lsense = 0;
  __syncthreads();
  cg::grid_group grid = cg::this_grid();

  long long local = 0;
  for (int it = 0; it < iters; ++it) {
    // Publish, barrier, then read a neighbor's slot: data must be visible.
    if (threadIdx.x == 0) {
      if (VARIANT == 2) {
        __hip_atomic_store(&data[blockIdx.x], it + (int)blockIdx.x,
                           __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      } else {
        data[blockIdx.x] = it + (int)blockIdx.x;
      }
    }
    if (VARIANT == 0) {
      grid.sync();
    } else {
      sense_barrier(count, sense, nblocks, &lsense, VARIANT == 1);
    }
    if (threadIdx.x == 0) {
      const int nb = ((int)blockIdx.x + 1) % nblocks;
      int v;
      if (VARIANT == 2) {
        v = __hip_atomic_load(&data[nb], __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_AGENT);
      } else {
        v = data[nb];
      }
      local += v;
    }
    if (VARIANT == 0) {
      grid.sync();
    } else {
      sense_barrier(count, sense, nblocks, &lsense, VARIANT == 1);
    }
  }
  if (threadIdx.x == 0) checksum[blockIdx.x] = local;
}

__global__ void empty_kernel(float* p) {
  if (p != nullptr) *p = 1.f;
}

}  // namespace

// Returns {seconds_total, checksum_sum, expected_sum}.
std::vector<double> barrier_probe(int64_t variant, int64_t wgs,
                                  int64_t iters) {
  auto opts = torch::TensorOptions()
                  .dtype(torch::kInt32)
                  .device(torch::kCUDA);
  auto count = torch::zeros({2}, opts);
  auto data = torch::zeros({wgs}, opts);
  auto checksum = torch::zeros({wgs}, opts.dtype(torch::kInt64));
  auto stream = at::cuda::getCurrentCUDAStream();

  int* count_p = count.data_ptr<int>();
  int* sense_p = count_p + 1;
  int* data_p = data.data_ptr<int>();
  long long* ck_p = (long long*)checksum.data_ptr<int64_t>();
  int iters_i = (int)iters, nblocks = (int)wgs;

  const void* kernels[3] = {(const void*)barrier_probe_kernel<0>,
                            (const void*)barrier_probe_kernel<1>,
                            (const void*)barrier_probe_kernel<2>};
  const void* kernel = kernels[variant];
  void* args[] = {&count_p, &sense_p, &data_p, &iters_i, &nblocks, &ck_p};

  // Warmup + timed (host wall clock around a synced region).
  DEVCHECK2(hipLaunchCooperativeKernel(kernel, dim3((uint32_t)wgs), dim3(256),
                                       args, 0, stream));
  DEVCHECK2(hipStreamSynchronize(stream));
  auto t0 = std::chrono::steady_clock::now();
  DEVCHECK2(hipLaunchCooperativeKernel(kernel, dim3((uint32_t)wgs), dim3(256),
                                       args, 0, stream));
  DEVCHECK2(hipStreamSynchronize(stream));
  auto t1 = std::chrono::steady_clock::now();

  double secs = std::chrono::duration<double>(t1 - t0).count();
  double got = (double)checksum.sum().item<int64_t>();
  // expected: sum over wg, it of (it + (wg+1)%wgs)
  double expected =
      (double)wgs * ((double)(iters - 1) * iters / 2.0) +
      (double)iters * ((double)(wgs - 1) * wgs / 2.0);
  return {secs, got, expected};
}

// Launch overhead: time `iters` empty kernel launches (for the hipGraph /
// persistent-kernel tradeoff).
double launch_probe(int64_t iters) {
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(empty_kernel, dim3(1), dim3(64), 0, stream, nullptr);
  DEVCHECK2(hipStreamSynchronize(stream));
  auto t0 = std::chrono::steady_clock::now();
  for (int64_t i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(empty_kernel, dim3(1), dim3(64), 0, stream, nullptr);
  }
  DEVCHECK2(hipStreamSynchronize(stream));
  auto t1 = std::chrono::steady_clock::now();
  return std::chrono::duration<double>(t1 - t0).count() / iters;
}

}  // namespace tbamd
