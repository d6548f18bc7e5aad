// Hand-written CDNA4 (gfx950 / MI355X) kernels for the IMPALA hot path.
//
// Design notes (see /root/repo/SURVEY.md §2.4 for the op inventory):
// - Shapes are small ([T=80, B=8..32, A~6..18], H≈256..520): on MI355X these
//   ops are LAUNCH- and LATENCY-bound, not FLOP-bound. The win over stock
//   PyTorch is fusing each algorithmic stage (V-trace: ~15 eager kernels +
//   an 80-iteration Python loop; LSTM: ~5 kernels × T steps × layers) into
//   ONE kernel, keeping every intermediate in registers/LDS.
// - V-trace's reverse-time scan is sequential per batch column: one
//   workgroup per column, wave-parallel over T for the log-prob phase
//   (ref math: torchbeast/core/vtrace.py:91-139).
// - The LSTM unroll is a persistent cooperative kernel: the T-step
//   recurrence runs inside one launch with grid.sync() between the gate
//   GEMV phase and the state-update phase, with the input GEMM
//   (x @ W_ih^T, the parallelizable 90% of the FLOPs) hoisted out to one
//   rocBLAS GEMM over all T (ref behavior: monobeast.py:599-611 —
//   done-masked state resets between steps).
// - All block sizes are multiples of the 64-wide wavefront.

#include <hip/hip_runtime.h>
#include <hip/hip_cooperative_groups.h>
#include <hiprand/hiprand_kernel.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <vector>

#include "atari_trunk.h"
#include "conv_mfma.h"

namespace tbamd {
std::vector<double> barrier_probe(int64_t variant, int64_t wgs, int64_t iters);
double launch_probe(int64_t iters);
}  // namespace tbamd (probes.hip)

namespace cg = cooperative_groups;

#define DEVCHECK(x) TORCH_CHECK(x == hipSuccess, "HIP error: ", hipGetErrorString(x))

static inline int ceil_div(int64_t a, int64_t b) { return (int)((a + b - 1) / b); }

// ---------------------------------------------------------------------------
// V-trace: log-probs + rho/c clip + reverse scan + pg advantages, one launch.
// ---------------------------------------------------------------------------

// One workgroup per batch column b. Dynamic LDS: [3*T] floats
// (cs, deltas, vs_minus_v).
__global__ void vtrace_kernel(
    const float* __restrict__ behavior_logits,  // [T,B,A]
    const float* __restrict__ target_logits,    // [T,B,A]
    const int64_t* __restrict__ actions,        // [T,B]
    const float* __restrict__ discounts,        // [T,B]
    const float* __restrict__ rewards,          // [T,B]
    const float* __restrict__ values,           // [T,B]
    const float* __restrict__ bootstrap,        // [B]
    float clip_rho, float clip_pg, int T, int B, int A,
    float* __restrict__ vs,        // [T,B]
    float* __restrict__ pg_adv,    // [T,B]
    float* __restrict__ log_rhos,  // [T,B]
    float* __restrict__ blp,       // [T,B]
    float* __restrict__ tlp) {     // [T,B]
  extern __shared__ float smem[];
  float* s_cs = smem;           // [T]
  float* s_delta = smem + T;    // [T]
  float* s_vmv = smem + 2 * T;  // [T]

  const int b = blockIdx.x;

  auto log_prob = [&](const float* logits, int t, int64_t a) {
    const float* row = logits + ((int64_t)t * B + b) * A;
    float m = row[0];
    for (int j = 1; j < A; ++j) m = fmaxf(m, row[j]);
    float s = 0.f;
    for (int j = 0; j < A; ++j) s += __expf(row[j] - m);
    return row[a] - m - __logf(s);
  };

  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    const int64_t idx = (int64_t)t * B + b;
    const int64_t a = actions[idx];
    const float t_lp = log_prob(target_logits, t, a);
    const float b_lp = log_prob(behavior_logits, t, a);
    const float lr = t_lp - b_lp;
    tlp[idx] = t_lp;
    blp[idx] = b_lp;
    log_rhos[idx] = lr;

    const float rho = __expf(lr);
    const float crho = fminf(rho, clip_rho);
    s_cs[t] = fminf(rho, 1.f);
    const float v_next = (t + 1 < T) ? values[idx + B] : bootstrap[b];
    s_delta[t] =
        crho * (rewards[idx] + discounts[idx] * v_next - values[idx]);
  }
  __syncthreads();

  if (threadIdx.x == 0) {
    // The inherently sequential reverse recurrence over T.
    float acc = 0.f;
    for (int t = T - 1; t >= 0; --t) {
      acc = s_delta[t] + discounts[(int64_t)t * B + b] * s_cs[t] * acc;
      s_vmv[t] = acc;
    }
  }
  __syncthreads();

  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    const int64_t idx = (int64_t)t * B + b;
    const float v = values[idx];
    vs[idx] = s_vmv[t] + v;
    const float vs_next =
        (t + 1 < T) ? (s_vmv[t + 1] + values[idx + B]) : bootstrap[b];
    const float rho = __expf(log_rhos[idx]);
    pg_adv[idx] = fminf(rho, clip_pg) *
                  (rewards[idx] + discounts[idx] * vs_next - v);
  }
}

std::vector<torch::Tensor> vtrace_from_logits(
    torch::Tensor behavior_logits, torch::Tensor target_logits,
    torch::Tensor actions, torch::Tensor discounts, torch::Tensor rewards,
    torch::Tensor values, torch::Tensor bootstrap, double clip_rho,
    double clip_pg) {
  TORCH_CHECK(behavior_logits.is_cuda(), "vtrace: expected GPU tensors");
  const int T = behavior_logits.size(0);
  const int B = behavior_logits.size(1);
  const int A = behavior_logits.size(2);
  auto opts = behavior_logits.options();
  auto vs = torch::empty({T, B}, opts);
  auto pg_adv = torch::empty({T, B}, opts);
  auto log_rhos = torch::empty({T, B}, opts);
  auto blp = torch::empty({T, B}, opts);
  auto tlp = torch::empty({T, B}, opts);
  auto actions_i64 = actions.to(torch::kInt64).contiguous();

  const int threads = 256;
  const size_t lds = 3 * (size_t)T * sizeof(float);
  hipLaunchKernelGGL(vtrace_kernel, dim3(B), dim3(threads), lds,
                     at::cuda::getCurrentCUDAStream(),
                     behavior_logits.data_ptr<float>(),
                     target_logits.data_ptr<float>(),
                     actions_i64.data_ptr<int64_t>(),
                     discounts.data_ptr<float>(), rewards.data_ptr<float>(),
                     values.data_ptr<float>(), bootstrap.data_ptr<float>(),
                     (float)clip_rho, (float)clip_pg, T, B, A,
                     vs.data_ptr<float>(), pg_adv.data_ptr<float>(),
                     log_rhos.data_ptr<float>(), blp.data_ptr<float>(),
                     tlp.data_ptr<float>());
  return {vs, pg_adv, log_rhos, blp, tlp};
}

// ---------------------------------------------------------------------------
// Fused IMPALA loss: pg + baseline + entropy values AND gradients, one pass.
// ---------------------------------------------------------------------------

// One thread per (t, b) row; block-reduced partial sums -> atomicAdd.
__global__ void impala_loss_kernel(
    const float* __restrict__ logits,     // [R, A], R = T*B
    const float* __restrict__ baseline,   // [R]
    const int64_t* __restrict__ actions,  // [R]
    const float* __restrict__ pg_advantages,  // [R]
    const float* __restrict__ vs,             // [R]
    int64_t R, int A,
    float* __restrict__ losses,        // [3]: pg, baseline, entropy
    float* __restrict__ d_logits_pg,   // [R, A]
    float* __restrict__ d_logits_ent,  // [R, A]
    float* __restrict__ d_baseline) {  // [R]
  __shared__ float red[3][256];

  const int64_t row = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  float pg_sum = 0.f, bl_sum = 0.f, ent_sum = 0.f;

  if (row < R) {
    const float* z = logits + row * A;
    float m = z[0];
    for (int j = 1; j < A; ++j) m = fmaxf(m, z[j]);
    float s = 0.f;
    for (int j = 0; j < A; ++j) s += __expf(z[j] - m);
    const float lse = m + __logf(s);
    const float inv_s = 1.f / s;

    // neg-entropy of this row: sum p*logp.
    float neg_ent = 0.f;
    for (int j = 0; j < A; ++j) {
      const float lp = z[j] - lse;
      neg_ent += __expf(z[j] - m) * inv_s * lp;
    }

    const int64_t a = actions[row];
    const float adv = pg_advantages[row];
    const float ce = lse - z[a];  // -log p[a]
    pg_sum = ce * adv;
    ent_sum = neg_ent;
    const float db = baseline[row] - vs[row];
    bl_sum = 0.5f * db * db;
    d_baseline[row] = db;

    for (int j = 0; j < A; ++j) {
      const float p = __expf(z[j] - m) * inv_s;
      const float lp = z[j] - lse;
      d_logits_pg[row * A + j] = adv * (p - (j == a ? 1.f : 0.f));
      d_logits_ent[row * A + j] = p * (lp - neg_ent);
    }
  }

  red[0][threadIdx.x] = pg_sum;
  red[1][threadIdx.x] = bl_sum;
  red[2][threadIdx.x] = ent_sum;
  __syncthreads();
  for (int stride = blockDim.x / 2; stride > 0; stride >>= 1) {
    if (threadIdx.x < stride) {
      red[0][threadIdx.x] += red[0][threadIdx.x + stride];
      red[1][threadIdx.x] += red[1][threadIdx.x + stride];
      red[2][threadIdx.x] += red[2][threadIdx.x + stride];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    atomicAdd(&losses[0], red[0][0]);
    atomicAdd(&losses[1], red[1][0]);
    atomicAdd(&losses[2], red[2][0]);
  }
}

std::vector<torch::Tensor> fused_impala_loss_fwd(
    torch::Tensor logits, torch::Tensor baseline, torch::Tensor actions,
    torch::Tensor pg_advantages, torch::Tensor vs) {
  TORCH_CHECK(logits.is_cuda(), "fused_impala_loss: expected GPU tensors");
  const int A = logits.size(-1);
  const int64_t R = logits.numel() / A;
  auto opts = logits.options();
  auto losses = torch::zeros({3}, opts);
  auto d_logits_pg = torch::empty_like(logits);
  auto d_logits_ent = torch::empty_like(logits);
  auto d_baseline = torch::empty_like(baseline);
  auto actions_i64 = actions.to(torch::kInt64).contiguous();

  const int threads = 256;
  hipLaunchKernelGGL(impala_loss_kernel, dim3(ceil_div(R, threads)),
                     dim3(threads), 0, at::cuda::getCurrentCUDAStream(),
                     logits.data_ptr<float>(), baseline.data_ptr<float>(),
                     actions_i64.data_ptr<int64_t>(),
                     pg_advantages.data_ptr<float>(), vs.data_ptr<float>(), R,
                     A, losses.data_ptr<float>(),
                     d_logits_pg.data_ptr<float>(),
                     d_logits_ent.data_ptr<float>(),
                     d_baseline.data_ptr<float>());
  return {losses[0], losses[1], losses[2], d_logits_pg, d_logits_ent,
          d_baseline};
}

// ---------------------------------------------------------------------------
// Fused global-norm clip + RMSProp over a flat parameter buffer.
// ---------------------------------------------------------------------------

__global__ void grad_norm_sq_kernel(const float* __restrict__ grad, int64_t n,
                                    float* __restrict__ out) {
  __shared__ float red[256];
  float acc = 0.f;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float g = grad[i];
    acc += g * g;
  }
  red[threadIdx.x] = acc;
  __syncthreads();
  for (int stride = blockDim.x / 2; stride > 0; stride >>= 1) {
    if (threadIdx.x < stride) red[threadIdx.x] += red[threadIdx.x + stride];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(out, red[0]);
}

__global__ void rmsprop_update_kernel(
    float* __restrict__ param, const float* __restrict__ grad,
    float* __restrict__ square_avg, int64_t n, float lr,
    const float* __restrict__ lr_ptr,  // overrides lr when non-null
    float alpha, float eps, float clip_norm,
    const float* __restrict__ norm_sq, float* __restrict__ norm_out) {
  const float norm = sqrtf(*norm_sq);
  float coef = 1.f;
  if (clip_norm > 0.f) coef = fminf(1.f, clip_norm / (norm + 1e-6f));
  if (blockIdx.x == 0 && threadIdx.x == 0) *norm_out = norm;
  // hipGraph capture freezes by-value arguments; when the learning rate
  // must keep decaying across graph replays it is read from device memory.
  if (lr_ptr != nullptr) lr = *lr_ptr;

  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const float g = grad[i] * coef;
    float s = square_avg[i];
    s = alpha * s + (1.f - alpha) * g * g;
    square_avg[i] = s;
    param[i] -= lr * g / (sqrtf(s) + eps);
  }
}

torch::Tensor rmsprop_step(torch::Tensor param, torch::Tensor grad,
                           torch::Tensor square_avg, double lr, double alpha,
                           double eps, double clip_norm,
                           std::optional<torch::Tensor> lr_tensor = {}) {
  TORCH_CHECK(param.is_cuda() && param.dim() == 1, "rmsprop: flat GPU tensor");
  const int64_t n = param.numel();
  auto norm_sq = torch::zeros({1}, param.options());
  auto norm_out = torch::empty({}, param.options());
  const float* lr_p =
      lr_tensor ? lr_tensor->data_ptr<float>() : nullptr;

  const int threads = 256;
  const int blocks = std::min<int64_t>(2048, ceil_div(n, threads));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(grad_norm_sq_kernel, dim3(blocks), dim3(threads), 0,
                     stream, grad.data_ptr<float>(), n,
                     norm_sq.data_ptr<float>());
  hipLaunchKernelGGL(rmsprop_update_kernel, dim3(blocks), dim3(threads), 0,
                     stream, param.data_ptr<float>(), grad.data_ptr<float>(),
                     square_avg.data_ptr<float>(), n, (float)lr, lr_p,
                     (float)alpha, (float)eps, (float)clip_norm,
                     norm_sq.data_ptr<float>(), norm_out.data_ptr<float>());
  return norm_out;
}

// ---------------------------------------------------------------------------
// Policy sampling: softmax + multinomial (philox) / argmax.
// ---------------------------------------------------------------------------

__global__ void policy_sample_kernel(const float* __restrict__ logits,
                                     int64_t N, int A, bool greedy,
                                     uint64_t seed,
                                     int64_t* __restrict__ actions) {
  const int64_t row = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  if (row >= N) return;
  const float* z = logits + row * A;

  if (greedy) {
    float best = z[0];
    int64_t best_j = 0;
    for (int j = 1; j < A; ++j) {
      if (z[j] > best) {
        best = z[j];
        best_j = j;
      }
    }
    actions[row] = best_j;
    return;
  }

  float m = z[0];
  for (int j = 1; j < A; ++j) m = fmaxf(m, z[j]);
  float s = 0.f;
  for (int j = 0; j < A; ++j) s += __expf(z[j] - m);

  hiprandStatePhilox4_32_10_t state;
  hiprand_init(seed, row, 0, &state);
  const float u = hiprand_uniform(&state) * s;

  float cdf = 0.f;
  int64_t pick = A - 1;
  for (int j = 0; j < A; ++j) {
    cdf += __expf(z[j] - m);
    if (u <= cdf) {
      pick = j;
      break;
    }
  }
  actions[row] = pick;
}

torch::Tensor policy_sample(torch::Tensor logits, bool greedy, int64_t seed) {
  TORCH_CHECK(logits.is_cuda(), "policy_sample: expected GPU tensor");
  const int A = logits.size(-1);
  const int64_t N = logits.numel() / A;
  auto out_sizes = logits.sizes().vec();
  out_sizes.pop_back();
  auto actions =
      torch::empty(out_sizes, logits.options().dtype(torch::kInt64));
  const int threads = 256;
  hipLaunchKernelGGL(policy_sample_kernel, dim3(ceil_div(N, threads)),
                     dim3(threads), 0, at::cuda::getCurrentCUDAStream(),
                     logits.data_ptr<float>(), N, A, greedy, (uint64_t)seed,
                     actions.data_ptr<int64_t>());
  return actions;
}

// ---------------------------------------------------------------------------
// Done-masked LSTM unroll (persistent cooperative kernel per layer).
// ---------------------------------------------------------------------------
// Done-masked LSTM unroll: v4 persistent kernels live in lstm_v4.hip.
// ---------------------------------------------------------------------------

namespace tbamd {
// v4 persistent-kernel launchers (lstm_v4.hip): h-slice ownership, LDS
// bf16 weight slices, agent-scope barriers (no cg::grid_sync).
struct Lstm4Geometry { int hs; int nblocks; size_t fwd_lds; size_t bwd_lds; };
Lstm4Geometry lstm4_geometry(int B, int H);
void lstm4_fwd_launch(torch::Tensor precomp, torch::Tensor w_hh_bf,
                      torch::Tensor notdone, torch::Tensor h0,
                      torch::Tensor c0, torch::Tensor out, torch::Tensor gates,
                      torch::Tensor hm, torch::Tensor cm,
                      torch::Tensor c_stash, torch::Tensor hT,
                      torch::Tensor cT);
void lstm4_bwd_launch(torch::Tensor gates, torch::Tensor cm,
                      torch::Tensor c_stash, torch::Tensor w_hh_t_bf,
                      torch::Tensor notdone, torch::Tensor d_out,
                      torch::Tensor d_hT, torch::Tensor d_cT,
                      torch::Tensor dgates, torch::Tensor dh_out,
                      torch::Tensor dc_out);

}  // namespace tbamd

// Multi-layer forward. flat_weights: [w_ih, w_hh, b_ih, b_hh] per layer.
std::vector<torch::Tensor> lstm_unroll_fwd(
    torch::Tensor x, torch::Tensor notdone, torch::Tensor h0,
    torch::Tensor c0, std::vector<torch::Tensor> flat_weights) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 3, "lstm: x must be [T,B,I] on GPU");
  const int T = x.size(0);
  const int B = x.size(1);
  const int L = flat_weights.size() / 4;
  const int H = h0.size(2);
  auto opts = x.options();
  auto stream = at::cuda::getCurrentCUDAStream();

  auto notdone_f = notdone.to(torch::kFloat32).contiguous();
  auto hT = torch::empty({L, B, H}, opts);
  auto cT = torch::empty({L, B, H}, opts);

  std::vector<torch::Tensor> stash;  // per layer: input, gates, hm, cm, c
  torch::Tensor layer_in = x.contiguous();
  torch::Tensor out;

  for (int l = 0; l < L; ++l) {
    auto w_ih = flat_weights[4 * l].contiguous();
    auto w_hh = flat_weights[4 * l + 1].contiguous();
    auto bias = (flat_weights[4 * l + 2] + flat_weights[4 * l + 3]);

    // The parallel 90%: one GEMM over all T*B rows.
    auto precomp =
        (torch::matmul(layer_in.reshape({(int64_t)T * B, -1}), w_ih.t()) +
         bias)
            .reshape({T, B, 4 * H})
            .contiguous();

    out = torch::empty({T, B, H}, opts);
    auto gates = torch::empty({T, B, 4 * H}, opts);
    auto hm = torch::empty({T, B, H}, opts);
    auto cm = torch::empty({T, B, H}, opts);
    auto c_out = torch::empty({T, B, H}, opts);
    auto h0_l = h0[l].contiguous();
    auto c0_l = c0[l].contiguous();

    auto w_hh_bf = w_hh.to(torch::kBFloat16).contiguous();
    tbamd::lstm4_fwd_launch(precomp, w_hh_bf, notdone_f, h0_l, c0_l, out,
                            gates, hm, cm, c_out, hT[l], cT[l]);

    stash.push_back(layer_in);
    stash.push_back(gates);
    stash.push_back(hm);
    stash.push_back(cm);
    stash.push_back(c_out);
    layer_in = out;
  }

  std::vector<torch::Tensor> result = {out, hT, cT};
  result.insert(result.end(), stash.begin(), stash.end());
  return result;
}

std::vector<torch::Tensor> lstm_unroll_bwd(
    torch::Tensor notdone, torch::Tensor h0, torch::Tensor c0,
    std::vector<torch::Tensor> flat_weights,
    std::vector<torch::Tensor> stash, torch::Tensor d_out_top,
    torch::Tensor d_hT, torch::Tensor d_cT) {
  const int L = flat_weights.size() / 4;
  const int T = d_out_top.size(0);
  const int B = d_out_top.size(1);
  const int H = h0.size(2);
  auto opts = d_out_top.options();
  auto stream = at::cuda::getCurrentCUDAStream();
  auto notdone_f = notdone.to(torch::kFloat32).contiguous();

  auto d_h0 = torch::empty({L, B, H}, opts);
  auto d_c0 = torch::empty({L, B, H}, opts);
  std::vector<torch::Tensor> d_weights(4 * L);

  torch::Tensor d_out = d_out_top.contiguous();
  for (int l = L - 1; l >= 0; --l) {
    auto layer_in = stash[5 * l];
    auto gates = stash[5 * l + 1];
    auto hm = stash[5 * l + 2];
    auto cm = stash[5 * l + 3];
    auto c_out = stash[5 * l + 4];
    auto w_ih = flat_weights[4 * l].contiguous();
    auto w_hh = flat_weights[4 * l + 1].contiguous();
    auto w_hh_t_bf =
        w_hh.t().contiguous().to(torch::kBFloat16).contiguous();  // [H,4H]

    auto dgates = torch::empty({T, B, 4 * H}, opts);
    auto dh = torch::empty({B, H}, opts);
    auto dc = torch::empty({B, H}, opts);
    auto d_hT_l = d_hT[l].contiguous();
    auto d_cT_l = d_cT[l].contiguous();
    tbamd::lstm4_bwd_launch(gates, cm, c_out, w_hh_t_bf, notdone_f, d_out,
                            d_hT_l, d_cT_l, dgates, dh, dc);

    // Weight/input grads: plain GEMMs (rocBLAS).
    auto dg2 = dgates.reshape({(int64_t)T * B, 4 * H});
    auto in2 = layer_in.reshape({(int64_t)T * B, -1});
    auto hm2 = hm.reshape({(int64_t)T * B, H});
    d_weights[4 * l] = torch::matmul(dg2.t(), in2);       // dW_ih
    d_weights[4 * l + 1] = torch::matmul(dg2.t(), hm2);   // dW_hh
    auto db = dg2.sum(0);
    d_weights[4 * l + 2] = db;                            // db_ih
    d_weights[4 * l + 3] = db.clone();                    // db_hh
    d_h0[l] = dh;
    d_c0[l] = dc;

    if (l > 0) {
      d_out = torch::matmul(dg2, w_ih).reshape({T, B, -1});
    } else {
      d_out = torch::matmul(dg2, w_ih).reshape({T, B, -1});
    }
  }

  std::vector<torch::Tensor> result = {d_out, d_h0, d_c0};
  result.insert(result.end(), d_weights.begin(), d_weights.end());
  return result;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("atari_trunk_fwd", &tbamd::atari_trunk_fwd);
  m.def("atari_trunk_supported", &tbamd::atari_trunk_supported);
  m.def("conv_trunk_fwd", &tbamd::conv_trunk_fwd);
  m.def("conv_trunk_mask_d3", &tbamd::conv_trunk_mask_d3);
  m.def("conv_trunk_dgrad3", &tbamd::conv_trunk_dgrad3);
  m.def("conv_trunk_dgrad2", &tbamd::conv_trunk_dgrad2);
  m.def("conv_trunk_wgrad1", &tbamd::conv_trunk_wgrad1);
  m.def("conv_trunk_wgrad2", &tbamd::conv_trunk_wgrad2);
  m.def("conv_trunk_wgrad3", &tbamd::conv_trunk_wgrad3);
  m.def("resnet_conv", &tbamd::resnet_conv);
  m.def("resnet_conv_supported", &tbamd::resnet_conv_supported);
  m.def("resnet_conv_wgrad", &tbamd::resnet_conv_wgrad);
  m.def("barrier_probe", &tbamd::barrier_probe);
  m.def("launch_probe", &tbamd::launch_probe);
  m.def("mfma_gemm", &tbamd::mfma_gemm);
  m.def("mfma_gemm_probe", &tbamd::mfma_gemm_probe);
  m.def("mfma_gemm_v2", &tbamd::mfma_gemm_v2);
  m.def("vtrace_from_logits", &vtrace_from_logits);
  m.def("fused_impala_loss_fwd", &fused_impala_loss_fwd);
  m.def("rmsprop_step", &rmsprop_step, py::arg("param"), py::arg("grad"),
        py::arg("square_avg"), py::arg("lr"), py::arg("alpha"),
        py::arg("eps"), py::arg("clip_norm"),
        py::arg("lr_tensor") = std::nullopt);
  m.def("policy_sample", &policy_sample);
  m.def("lstm_unroll_fwd", &lstm_unroll_fwd);
  m.def("lstm_unroll_bwd", &lstm_unroll_bwd);
}
