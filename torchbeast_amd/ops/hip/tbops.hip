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
    float* __restrict__ square_avg, int64_t n, float lr, float alpha,
    float eps, float clip_norm, const float* __restrict__ norm_sq,
    float* __restrict__ norm_out) {
  const float norm = sqrtf(*norm_sq);
  float coef = 1.f;
  if (clip_norm > 0.f) coef = fminf(1.f, clip_norm / (norm + 1e-6f));
  if (blockIdx.x == 0 && threadIdx.x == 0) *norm_out = norm;

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
                           double eps, double clip_norm) {
  TORCH_CHECK(param.is_cuda() && param.dim() == 1, "rmsprop: flat GPU tensor");
  const int64_t n = param.numel();
  auto norm_sq = torch::zeros({1}, param.options());
  auto norm_out = torch::empty({}, param.options());

  const int threads = 256;
  const int blocks = std::min<int64_t>(2048, ceil_div(n, threads));
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(grad_norm_sq_kernel, dim3(blocks), dim3(threads), 0,
                     stream, grad.data_ptr<float>(), n,
                     norm_sq.data_ptr<float>());
  hipLaunchKernelGGL(rmsprop_update_kernel, dim3(blocks), dim3(threads), 0,
                     stream, param.data_ptr<float>(), grad.data_ptr<float>(),
                     square_avg.data_ptr<float>(), n, (float)lr, (float)alpha,
                     (float)eps, (float)clip_norm,
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

// Forward, one layer, persistent cooperative kernel. Each workgroup OWNS a
// slice of W_hh rows staged in LDS for the whole unroll (one load instead of
// T re-reads from L3 — W_hh is ~4 MB for hidden 519, larger than one XCD's
// L2). Per step: gate phase (each WG computes its gate slice for all B) ->
// grid.sync -> state-update phase (grid-stride over B*H, which also writes
// the next step's done-masked state) -> grid.sync.
__global__ __launch_bounds__(256) void lstm_fwd_kernel(
    const float* __restrict__ precomp,  // [T,B,4H] = x@W_ih^T + b_ih + b_hh
    const float* __restrict__ w_hh,     // [4H,H]
    const float* __restrict__ notdone,  // [T,B]
    const float* __restrict__ h0,       // [B,H]
    const float* __restrict__ c0,       // [B,H]
    int T, int B, int H, int j_slice,
    float* __restrict__ out,    // [T,B,H]
    float* __restrict__ gates,  // [T,B,4H] post-activation
    float* __restrict__ hm,     // [T,B,H] masked h_{t-1}
    float* __restrict__ cm,     // [T,B,H] masked c_{t-1}
    float* __restrict__ c_out,  // [T,B,H]
    float* __restrict__ hT,     // [B,H]
    float* __restrict__ cT) {   // [B,H]
  cg::grid_group grid = cg::this_grid();
  extern __shared__ float s_w[];  // [j_slice, H]
  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t nthreads = (int64_t)gridDim.x * blockDim.x;
  const int64_t BH = (int64_t)B * H;
  const int G4 = 4 * H;

  const int j0 = blockIdx.x * j_slice;
  const int jn = min(j_slice, G4 - j0);

  // Stage this workgroup's W_hh rows into LDS (once for the whole unroll).
  for (int i = threadIdx.x; i < jn * H; i += blockDim.x) {
    s_w[i] = w_hh[(int64_t)j0 * H + i];
  }

  // Initialize hm/cm for t=0.
  for (int64_t i = tid; i < BH; i += nthreads) {
    const int b = i / H;
    const float nd = notdone[b];
    hm[i] = nd * h0[i];
    cm[i] = nd * c0[i];
  }
  grid.sync();

  for (int t = 0; t < T; ++t) {
    const float* hm_t = hm + (int64_t)t * BH;
    const float* cm_t = cm + (int64_t)t * BH;
    float* gates_t = gates + (int64_t)t * (int64_t)B * G4;
    const float* pre_t = precomp + (int64_t)t * (int64_t)B * G4;

    // Gate phase: this WG computes gates[b, j0..j0+jn) for all b.
    for (int i = threadIdx.x; i < jn * B; i += blockDim.x) {
      const int b = i / jn;           // lanes iterate j fastest: the hm row
      const int jl = i - (i / jn) * jn;  // is shared across the wave.
      const float* hrow = hm_t + (int64_t)b * H;
      const float* wrow = s_w + (int64_t)jl * H;
      float acc = pre_t[(int64_t)b * G4 + j0 + jl];
      for (int h = 0; h < H; ++h) acc += hrow[h] * wrow[h];
      const int gate = (j0 + jl) / H;
      if (gate == 2) {
        acc = tanhf(acc);
      } else {
        acc = 1.f / (1.f + __expf(-acc));
      }
      gates_t[(int64_t)b * G4 + j0 + jl] = acc;
    }
    grid.sync();

    // State update (+ prepare next step's masked state).
    float* out_t = out + (int64_t)t * BH;
    float* c_t = c_out + (int64_t)t * BH;
    for (int64_t i = tid; i < BH; i += nthreads) {
      const int b = i / H;
      const int h = i - (int64_t)(i / H) * H;
      const float* g4 = gates_t + (int64_t)b * G4;
      const float gi = g4[h];
      const float gf = g4[H + h];
      const float gg = g4[2 * H + h];
      const float go = g4[3 * H + h];
      const float c_new = gf * cm_t[i] + gi * gg;
      const float h_new = go * tanhf(c_new);
      c_t[i] = c_new;
      out_t[i] = h_new;
      if (t + 1 < T) {
        const float nd = notdone[(int64_t)(t + 1) * B + b];
        hm[(int64_t)(t + 1) * BH + i] = nd * h_new;
        cm[(int64_t)(t + 1) * BH + i] = nd * c_new;
      } else {
        hT[i] = h_new;
        cT[i] = c_new;
      }
    }
    grid.sync();
  }
}

// Backward, one layer: computes pre-activation gate grads dgates [T,B,4H]
// and the carried dh/dc. Phase 1 (elementwise over B*H) produces dgates;
// phase 2 computes dh_{t-1} = dgates_t @ W_hh using a per-workgroup LDS
// slice of W_hh^T held across all T steps (same persistent-weights design
// as the forward). Weight/input grads are batched rocBLAS GEMMs on the host.
__global__ __launch_bounds__(256) void lstm_bwd_kernel(
    const float* __restrict__ gates,    // [T,B,4H] post-activation
    const float* __restrict__ cm,       // [T,B,H]
    const float* __restrict__ c_out,    // [T,B,H]
    const float* __restrict__ w_hh_t,   // [H,4H] = W_hh^T (contiguous)
    const float* __restrict__ notdone,  // [T,B]
    const float* __restrict__ d_out,    // [T,B,H]
    const float* __restrict__ d_hT,     // [B,H]
    const float* __restrict__ d_cT,     // [B,H]
    int T, int B, int H, int h_slice,
    float* __restrict__ dgates,  // [T,B,4H] PRE-activation grads
    float* __restrict__ dh,      // [B,H] workspace, ends as d_h0
    float* __restrict__ dc) {    // [B,H] workspace, ends as d_c0
  cg::grid_group grid = cg::this_grid();
  extern __shared__ float s_wt[];  // [h_slice, 4H]
  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t nthreads = (int64_t)gridDim.x * blockDim.x;
  const int64_t BH = (int64_t)B * H;
  const int G4 = 4 * H;

  const int hh0 = blockIdx.x * h_slice;
  const int hn = min(h_slice, H - hh0);

  for (int i = threadIdx.x; i < hn * G4; i += blockDim.x) {
    s_wt[i] = w_hh_t[(int64_t)hh0 * G4 + i];
  }

  for (int64_t i = tid; i < BH; i += nthreads) {
    dh[i] = d_hT[i];
    dc[i] = d_cT[i];
  }
  grid.sync();

  for (int t = T - 1; t >= 0; --t) {
    const float* gates_t = gates + (int64_t)t * (int64_t)B * G4;
    const float* cm_t = cm + (int64_t)t * BH;
    const float* c_t = c_out + (int64_t)t * BH;
    float* dgates_t = dgates + (int64_t)t * (int64_t)B * G4;

    // Phase 1: per-(b,h) gate gradients; dc becomes masked d c_{t-1}.
    for (int64_t i = tid; i < BH; i += nthreads) {
      const int b = i / H;
      const int h = i - (int64_t)(i / H) * H;
      const float* g4 = gates_t + (int64_t)b * G4;
      const float gi = g4[h];
      const float gf = g4[H + h];
      const float gg = g4[2 * H + h];
      const float go = g4[3 * H + h];

      const float dh_t = dh[i] + d_out[(int64_t)t * BH + i];
      const float tc = tanhf(c_t[i]);
      float dc_t = dc[i] + dh_t * go * (1.f - tc * tc);

      const float d_go = dh_t * tc;
      const float d_gi = dc_t * gg;
      const float d_gf = dc_t * cm_t[i];
      const float d_gg = dc_t * gi;

      float* dg4 = dgates_t + (int64_t)b * G4;
      dg4[h] = d_gi * gi * (1.f - gi);
      dg4[H + h] = d_gf * gf * (1.f - gf);
      dg4[2 * H + h] = d_gg * (1.f - gg * gg);
      dg4[3 * H + h] = d_go * go * (1.f - go);

      const float nd = notdone[(int64_t)t * B + b];
      dc[i] = nd * dc_t * gf;
    }
    grid.sync();

    // Phase 2: dh_{t-1}[b, h] for this WG's h-slice, from the LDS W_hh^T.
    for (int i = threadIdx.x; i < hn * B; i += blockDim.x) {
      const int b = i / hn;
      const int hl = i - (i / hn) * hn;
      const float* dg4 = dgates_t + (int64_t)b * G4;
      const float* wrow = s_wt + (int64_t)hl * G4;
      float acc = 0.f;
      for (int j = 0; j < G4; ++j) acc += dg4[j] * wrow[j];
      const float nd = notdone[(int64_t)t * B + b];
      dh[(int64_t)b * H + hh0 + hl] = nd * acc;
    }
    grid.sync();
  }
}

static int coop_grid_size(const void* kernel, int threads, size_t lds) {
  int device;
  DEVCHECK(hipGetDevice(&device));
  hipDeviceProp_t props;
  DEVCHECK(hipGetDeviceProperties(&props, device));
  int blocks_per_cu = 0;
  DEVCHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &blocks_per_cu, kernel, threads, lds));
  return std::max(1, blocks_per_cu * props.multiProcessorCount);
}

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

  const int threads = 256;
  // Workgroup count: each WG owns a W_hh row-slice held in LDS. Cap the
  // per-WG slice at ~48 KiB so several blocks fit per CU.
  const int64_t w_bytes = (int64_t)4 * H * H * 4;
  int grid = std::max<int64_t>(128, ceil_div(w_bytes, 48 * 1024));
  const int j_slice = ceil_div((int64_t)4 * H, grid);
  const size_t fwd_lds = (size_t)j_slice * H * sizeof(float);
  TORCH_CHECK(fwd_lds <= 64 * 1024, "lstm hidden size too large for LDS slices");
  {
    int cap = coop_grid_size((const void*)lstm_fwd_kernel, threads, fwd_lds);
    TORCH_CHECK(grid <= cap, "lstm fwd grid exceeds cooperative capacity");
  }

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

    const float* precomp_p = precomp.data_ptr<float>();
    const float* w_hh_p = w_hh.data_ptr<float>();
    const float* notdone_p = notdone_f.data_ptr<float>();
    const float* h0_p = h0_l.data_ptr<float>();
    const float* c0_p = c0_l.data_ptr<float>();
    float* out_p = out.data_ptr<float>();
    float* gates_p = gates.data_ptr<float>();
    float* hm_p = hm.data_ptr<float>();
    float* cm_p = cm.data_ptr<float>();
    float* c_out_p = c_out.data_ptr<float>();
    float* hT_p = hT[l].data_ptr<float>();
    float* cT_p = cT[l].data_ptr<float>();
    int T_ = T, B_ = B, H_ = H, j_slice_ = j_slice;
    void* args[] = {&precomp_p, &w_hh_p, &notdone_p, &h0_p, &c0_p,
                    &T_,        &B_,     &H_,        &j_slice_,
                    &out_p,     &gates_p, &hm_p,     &cm_p,
                    &c_out_p,   &hT_p,   &cT_p};
    DEVCHECK(hipLaunchCooperativeKernel((const void*)lstm_fwd_kernel,
                                        dim3(grid), dim3(threads), args,
                                        fwd_lds, stream));

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

  const int threads = 256;
  const int64_t w_bytes = (int64_t)4 * H * H * 4;
  int grid = std::max<int64_t>(128, ceil_div(w_bytes, 48 * 1024));
  const int h_slice = ceil_div((int64_t)H, grid);
  const size_t bwd_lds = (size_t)h_slice * 4 * H * sizeof(float);
  TORCH_CHECK(bwd_lds <= 64 * 1024, "lstm hidden size too large for LDS slices");
  {
    int cap = coop_grid_size((const void*)lstm_bwd_kernel, threads, bwd_lds);
    TORCH_CHECK(grid <= cap, "lstm bwd grid exceeds cooperative capacity");
  }

  torch::Tensor d_out = d_out_top.contiguous();
  for (int l = L - 1; l >= 0; --l) {
    auto layer_in = stash[5 * l];
    auto gates = stash[5 * l + 1];
    auto hm = stash[5 * l + 2];
    auto cm = stash[5 * l + 3];
    auto c_out = stash[5 * l + 4];
    auto w_ih = flat_weights[4 * l].contiguous();
    auto w_hh = flat_weights[4 * l + 1].contiguous();
    auto w_hh_t = w_hh.t().contiguous();  // [H, 4H] for the bwd LDS slices

    auto dgates = torch::empty({T, B, 4 * H}, opts);
    auto dh = torch::empty({B, H}, opts);
    auto dc = torch::empty({B, H}, opts);
    auto d_hT_l = d_hT[l].contiguous();
    auto d_cT_l = d_cT[l].contiguous();

    const float* gates_p = gates.data_ptr<float>();
    const float* cm_p = cm.data_ptr<float>();
    const float* c_out_p = c_out.data_ptr<float>();
    const float* w_hh_t_p = w_hh_t.data_ptr<float>();
    const float* notdone_p = notdone_f.data_ptr<float>();
    const float* d_out_p = d_out.data_ptr<float>();
    const float* d_hT_p = d_hT_l.data_ptr<float>();
    const float* d_cT_p = d_cT_l.data_ptr<float>();
    float* dgates_p = dgates.data_ptr<float>();
    float* dh_p = dh.data_ptr<float>();
    float* dc_p = dc.data_ptr<float>();
    int T_ = T, B_ = B, H_ = H, h_slice_ = h_slice;
    void* args[] = {&gates_p, &cm_p, &c_out_p, &w_hh_t_p, &notdone_p,
                    &d_out_p, &d_hT_p, &d_cT_p, &T_, &B_, &H_, &h_slice_,
                    &dgates_p, &dh_p, &dc_p};
    DEVCHECK(hipLaunchCooperativeKernel((const void*)lstm_bwd_kernel,
                                        dim3(grid), dim3(threads), args,
                                        bwd_lds, stream));

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
  m.def("barrier_probe", &tbamd::barrier_probe);
  m.def("launch_probe", &tbamd::launch_probe);
  m.def("mfma_gemm", &tbamd::mfma_gemm);
  m.def("mfma_gemm_probe", &tbamd::mfma_gemm_probe);
  m.def("mfma_gemm_v2", &tbamd::mfma_gemm_v2);
  m.def("vtrace_from_logits", &vtrace_from_logits);
  m.def("fused_impala_loss_fwd", &fused_impala_loss_fwd);
  m.def("rmsprop_step", &rmsprop_step);
  m.def("policy_sample", &policy_sample);
  m.def("lstm_unroll_fwd", &lstm_unroll_fwd);
  m.def("lstm_unroll_bwd", &lstm_unroll_bwd);
}
