// Done-masked LSTM unroll v4 for gfx950 — persistent kernels without
// cooperative-groups grid.sync.
//
// Why: the v2 design used cg::grid_group::sync between the gate and
// state-update phases. Measured on MI355X (profiles/PROFILE_r2.md), each
// sync's device-scope release/acquire invalidates every XCD's L2, so all
// per-step state (precomp, gates, h, c) re-misses after every one of the
// 2*T syncs: 13.6 ms per [T=80,B=32,H=519] layer forward (~170 us/step for
// ~5 us of math).
//
// v4 design (MI355X-first):
// * h-slice ownership: workgroup g owns hidden rows [g*hs, g*hs+hs) and
//   computes ALL FOUR gates and the c/h update for its rows. The cell
//   state c never leaves the workgroup (LDS-resident across all T steps);
//   only h crosses workgroups.
// * Cross-workgroup traffic (h forward, dgates/dh backward) flows through
//   agent-scope relaxed atomics — individually coherent at the coherence
//   point, so the barrier needs NO cache-flushing fence.
// * The barrier is a sense-reversing counter of agent-scope atomics with
//   s_sleep backoff: measured 2.7-6.8 us at 32-128 workgroups vs
//   7.6-25 us for cg (scripts/probe_barrier.py), and critically it leaves
//   the L2s warm. One barrier per forward step, two per backward step.
// * W_hh lives in LDS as bf16 row-slices staged once for the whole unroll
//   (4*hs rows for forward; the transposed hs columns for backward).
//   Gate dots are bf16 x bf16 -> fp32; the c-state path stays fp32.
// * Launched via hipLaunchCooperativeKernel purely for the co-residency
//   guarantee (grid <= 2 blocks/CU worth of LDS); grid.sync is never used.
//
// Reference behavior being implemented: the per-step done-masked unroll of
// torchbeast/monobeast.py:599-611 (state <- notdone*state, then one LSTM
// step), including the [input, gates, hm, cm, c] stash contract the host
// GEMMs (dW_ih, dW_hh, dx) consume.

#include <ATen/cuda/CUDAContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <cstdlib>
#include <vector>

namespace tbamd {

namespace {

constexpr int kNT = 512;  // 8 waves

using bf16x8v = __attribute__((ext_vector_type(8))) __bf16;

__device__ __forceinline__ void agent_barrier(int* count, int* sense,
                                              int nblocks, int* lsense) {
  __syncthreads();
  if (threadIdx.x == 0) {
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
  }
  __syncthreads();
}

__device__ __forceinline__ float scoped_load(const float* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ void scoped_store(float* p, float v) {
  __hip_atomic_store(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

__device__ __forceinline__ float bf2f(__bf16 v) { return (float)v; }

// ---------------------------------------------------------------------------
// Forward. One barrier per step.
// LDS layout (dynamic): sW bf16 [4*hs][H] | s_hm bf16 [B][H] |
//                       s_gates f32 [4*hs*B] | c_local f32 [B*hs]
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(kNT) void lstm4_fwd_kernel(
    const float* __restrict__ precomp,  // [T,B,4H] = x@W_ih^T + b_ih + b_hh
    const __bf16* __restrict__ w_hh,    // [4H,H] bf16
    const float* __restrict__ notdone,  // [T,B]
    const float* __restrict__ c0,       // [B,H]
    int T, int B, int H, int hs, int nblocks,
    float* __restrict__ hglob,  // [2,B*H]; buf 0 prefilled with h0
    int* __restrict__ bar,      // {count, sense}
    float* __restrict__ out,    // [T,B,H]
    float* __restrict__ gates,  // [T,B,4H] post-activation (stash)
    float* __restrict__ hm,     // [T,B,H] masked h_{t-1} (stash)
    float* __restrict__ cm,     // [T,B,H] masked c_{t-1} (stash)
    float* __restrict__ c_out,  // [T,B,H] (stash)
    float* __restrict__ hT, float* __restrict__ cT) {  // [B,H]
  const int HP = (H + 7) & ~7;  // LDS rows padded to 16 B for b128 reads
  extern __shared__ char smem[];
  __bf16* sW = reinterpret_cast<__bf16*>(smem);               // [4*hs][HP]
  __bf16* s_hm = sW + (size_t)4 * hs * HP;                    // [B][HP]
  float* s_gates = reinterpret_cast<float*>(s_hm + (size_t)B * HP);  // [4*hs*B]
  float* c_local = s_gates + (size_t)4 * hs * B;              // [B*hs]
  __shared__ int lsense;
  if (threadIdx.x == 0) lsense = 0;

  const int tid = threadIdx.x;
  const int r0 = blockIdx.x * hs;
  const int rs = min(hs, H - r0);  // may be <= 0 for trailing blocks
  const int64_t BH = (int64_t)B * H;
  const int G4 = 4 * H;

  // Stage this workgroup's W_hh rows (gate-major slices) once; zero the
  // row pad so vectorized dots read zeros past H.
  if (rs > 0) {
    for (int i = tid; i < 4 * rs * HP; i += kNT) {
      const int row = i / HP;          // g*rs + r
      const int g = row / rs, r = row % rs;
      const int h = i % HP;
      sW[(g * hs + r) * HP + h] =
          (h < H) ? w_hh[(int64_t)(g * H + r0 + r) * H + h] : (__bf16)0.f;
    }
    for (int i = tid; i < B * rs; i += kNT) {
      const int b = i / rs, r = i % rs;
      c_local[b * hs + r] = c0[(int64_t)b * H + r0 + r];
    }
  }
  // Zero the s_hm row pads once (stage only writes h < H).
  for (int b = 0; b < B; ++b) {
    for (int h = H + tid; h < HP; h += kNT) s_hm[(size_t)b * HP + h] = (__bf16)0.f;
  }
  __syncthreads();

  int buf = 0;
  for (int t = 0; t < T; ++t) {
    const float* nd_t = notdone + (int64_t)t * B;
    // ---- stage masked h into LDS (and the hm stash, written by block 0).
    // 8-deep manual prefetch: scoped loads bypass L2, so un-pipelined
    // serial loads would each pay the full coherence-point latency.
    {
      const float* src = hglob + (int64_t)buf * BH;
      float* hm_t = hm + (int64_t)t * BH;
      const int total = (int)BH;
      for (int base = tid * 8; base < total; base += kNT * 8) {
        float v[8];
        const int n = min(8, total - base);
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          if (u < n) v[u] = scoped_load(&src[base + u]);
        }
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          if (u < n) {
            const int i = base + u;
            const float m = v[u] * nd_t[i / H];
            s_hm[(size_t)(i / H) * HP + i % H] = (__bf16)m;
            if (blockIdx.x == 0) hm_t[i] = m;
          }
        }
      }
    }
    __syncthreads();

    // ---- gate phase: all 4 gates for this WG's rows, all b.
    if (rs > 0) {
      const float* pre_t = precomp + (int64_t)t * B * G4;
      float* gates_t = gates + (int64_t)t * B * G4;
      const int jn4 = 4 * rs;
      for (int i = tid; i < B * jn4; i += kNT) {
        const int b = i / jn4;
        const int jl = i % jn4;
        const int g = jl / rs, r = jl % rs;
        const __bf16* wrow = sW + (size_t)(g * hs + r) * HP;
        const __bf16* hrow = s_hm + (size_t)b * HP;
        float acc = pre_t[(int64_t)b * G4 + g * H + r0 + r];
        // Vectorized LDS reads (b128): rows are 16 B-padded; pads are zero.
        for (int h8 = 0; h8 < HP; h8 += 8) {
          const bf16x8v hv = *reinterpret_cast<const bf16x8v*>(&hrow[h8]);
          const bf16x8v wv = *reinterpret_cast<const bf16x8v*>(&wrow[h8]);
#pragma unroll
          for (int u = 0; u < 8; ++u) acc += bf2f(hv[u]) * bf2f(wv[u]);
        }
        acc = (g == 2) ? tanhf(acc) : 1.f / (1.f + __expf(-acc));
        s_gates[(g * hs + r) * B + b] = acc;
        gates_t[(int64_t)b * G4 + g * H + r0 + r] = acc;
      }
    }
    __syncthreads();

    // ---- state update for this WG's rows.
    if (rs > 0) {
      float* out_t = out + (int64_t)t * BH;
      float* cm_t = cm + (int64_t)t * BH;
      float* c_t = c_out + (int64_t)t * BH;
      float* dst = hglob + (int64_t)(buf ^ 1) * BH;
      for (int i = tid; i < B * rs; i += kNT) {
        const int b = i / rs, r = i % rs;
        const int64_t gi_idx = (int64_t)b * H + r0 + r;
        const float gi = s_gates[(0 * hs + r) * B + b];
        const float gf = s_gates[(1 * hs + r) * B + b];
        const float gg = s_gates[(2 * hs + r) * B + b];
        const float go = s_gates[(3 * hs + r) * B + b];
        const float c_m = c_local[b * hs + r] * nd_t[b];
        const float c_new = gf * c_m + gi * gg;
        const float h_new = go * tanhf(c_new);
        c_local[b * hs + r] = c_new;
        cm_t[gi_idx] = c_m;
        c_t[gi_idx] = c_new;
        out_t[gi_idx] = h_new;
        scoped_store(&dst[gi_idx], h_new);
        if (t == T - 1) {
          hT[gi_idx] = h_new;
          cT[gi_idx] = c_new;
        }
      }
    }
    agent_barrier(bar, bar + 1, nblocks, &lsense);
    buf ^= 1;
  }
}

// ---------------------------------------------------------------------------
// Backward. Two barriers per step (dgates publish, dh consume).
// LDS: sWT bf16 [hs][4H] | s_dg bf16 [DB][4H] | s_part f32 [DB*hs*SEG] |
//      dc_local f32 [B*hs]
// ---------------------------------------------------------------------------

constexpr int kDB = 8;   // dgates batch-chunk staged per round
constexpr int kSEG = 8;  // per-dot K-split for thread utilization

__global__ __launch_bounds__(kNT) void lstm4_bwd_kernel(
    const float* __restrict__ gates,    // [T,B,4H] post-activation
    const float* __restrict__ cm,       // [T,B,H]
    const float* __restrict__ c_out,    // [T,B,H]
    const __bf16* __restrict__ w_hh_t,  // [H,4H] bf16 (W_hh transposed)
    const float* __restrict__ notdone,  // [T,B]
    const float* __restrict__ d_out,    // [T,B,H]
    const float* __restrict__ d_cT,     // [B,H]
    int T, int B, int H, int hs, int nblocks,
    float* __restrict__ dhglob,        // [2,B*H]; buf 0 prefilled d_hT
    float* __restrict__ dgates_comm,   // [B,4H] scoped scratch
    int* __restrict__ bar,
    float* __restrict__ dgates,  // [T,B,4H] PRE-activation grads (stash)
    float* __restrict__ dh_out, float* __restrict__ dc_out) {  // [B,H]
  const int G4P = (4 * H + 7) & ~7;  // padded rows for b128 reads
  extern __shared__ char smem[];
  __bf16* sWT = reinterpret_cast<__bf16*>(smem);          // [hs][G4P]
  __bf16* s_dg = sWT + (size_t)hs * G4P;                  // [kDB][G4P]
  float* s_part =
      reinterpret_cast<float*>(s_dg + (size_t)kDB * G4P);  // [kDB*hs*kSEG]
  float* dc_local = s_part + (size_t)kDB * hs * kSEG;      // [B*hs]
  __shared__ int lsense;
  if (threadIdx.x == 0) lsense = 0;

  const int tid = threadIdx.x;
  const int r0 = blockIdx.x * hs;
  const int rs = min(hs, H - r0);
  const int64_t BH = (int64_t)B * H;
  const int G4 = 4 * H;

  if (rs > 0) {
    for (int i = tid; i < rs * G4P; i += kNT) {
      const int r = i / G4P;
      const int j = i % G4P;
      sWT[(size_t)r * G4P + j] =
          (j < G4) ? w_hh_t[(int64_t)(r0 + r) * G4 + j] : (__bf16)0.f;
    }
    for (int i = tid; i < B * rs; i += kNT) {
      const int b = i / rs, r = i % rs;
      dc_local[b * hs + r] = d_cT[(int64_t)b * H + r0 + r];
    }
  }
  // Zero the s_dg row pads once (chunk staging only writes j < G4).
  for (int b = 0; b < kDB; ++b) {
    for (int j = G4 + tid; j < G4P; j += kNT) {
      s_dg[(size_t)b * G4P + j] = (__bf16)0.f;
    }
  }
  __syncthreads();

  int buf = 0;
  for (int t = T - 1; t >= 0; --t) {
    const float nd_t0 = 0.f;  // silence unused warnings in rs==0 blocks
    (void)nd_t0;
    // ---- phase 1: pre-activation gate grads for this WG's rows.
    if (rs > 0) {
      const float* gates_t = gates + (int64_t)t * B * G4;
      const float* cm_t = cm + (int64_t)t * BH;
      const float* c_t = c_out + (int64_t)t * BH;
      const float* dout_t = d_out + (int64_t)t * BH;
      float* dg_t = dgates + (int64_t)t * B * G4;
      const float* src = dhglob + (int64_t)buf * BH;
      for (int i = tid; i < B * rs; i += kNT) {
        const int b = i / rs, r = i % rs;
        const int64_t hi = (int64_t)b * H + r0 + r;
        const float gi = gates_t[(int64_t)b * G4 + 0 * H + r0 + r];
        const float gf = gates_t[(int64_t)b * G4 + 1 * H + r0 + r];
        const float gg = gates_t[(int64_t)b * G4 + 2 * H + r0 + r];
        const float go = gates_t[(int64_t)b * G4 + 3 * H + r0 + r];

        const float dh_t = scoped_load(&src[hi]) + dout_t[hi];
        const float tc = tanhf(c_t[hi]);
        const float dc_t = dc_local[b * hs + r] + dh_t * go * (1.f - tc * tc);

        const float d_gi = (dc_t * gg) * gi * (1.f - gi);
        const float d_gf = (dc_t * cm_t[hi]) * gf * (1.f - gf);
        const float d_gg = (dc_t * gi) * (1.f - gg * gg);
        const float d_go = (dh_t * tc) * go * (1.f - go);

        dg_t[(int64_t)b * G4 + 0 * H + r0 + r] = d_gi;
        dg_t[(int64_t)b * G4 + 1 * H + r0 + r] = d_gf;
        dg_t[(int64_t)b * G4 + 2 * H + r0 + r] = d_gg;
        dg_t[(int64_t)b * G4 + 3 * H + r0 + r] = d_go;
        scoped_store(&dgates_comm[(int64_t)b * G4 + 0 * H + r0 + r], d_gi);
        scoped_store(&dgates_comm[(int64_t)b * G4 + 1 * H + r0 + r], d_gf);
        scoped_store(&dgates_comm[(int64_t)b * G4 + 2 * H + r0 + r], d_gg);
        scoped_store(&dgates_comm[(int64_t)b * G4 + 3 * H + r0 + r], d_go);

        const float nd = notdone[(int64_t)t * B + b];
        dc_local[b * hs + r] = nd * dc_t * gf;
      }
    }
    agent_barrier(bar, bar + 1, nblocks, &lsense);

    // ---- phase 2: dh_{t-1}[b][h'] = nd * sum_j dgates[b][j] * W_hh[j][h']
    // for this WG's h'-rows, chunked over b with a per-dot K-split.
    {
      float* dst = dhglob + (int64_t)(buf ^ 1) * BH;
      // Segment bounds are 8-aligned so segment dots use b128 reads.
      const int seg_len = ((G4P / 8 + kSEG - 1) / kSEG) * 8;
      for (int b0 = 0; b0 < B; b0 += kDB) {
        const int db = min(kDB, B - b0);
        {
          const int total = db * G4;
          for (int base = tid * 8; base < total; base += kNT * 8) {
            float v[8];
            const int n = min(8, total - base);
#pragma unroll
            for (int u = 0; u < 8; ++u) {
              if (u < n) {
                const int i = base + u;
                v[u] = scoped_load(
                    &dgates_comm[(int64_t)(b0 + i / G4) * G4 + i % G4]);
              }
            }
#pragma unroll
            for (int u = 0; u < 8; ++u) {
              if (u < n) {
                const int i = base + u;
                s_dg[(size_t)(i / G4) * G4P + i % G4] = (__bf16)v[u];
              }
            }
          }
        }
        __syncthreads();
        if (rs > 0) {
          for (int u = tid; u < db * rs * kSEG; u += kNT) {
            const int seg = u % kSEG;
            const int r = (u / kSEG) % rs;
            const int bl = u / (kSEG * rs);
            const int j0 = seg * seg_len;
            const int j1 = min(G4P, j0 + seg_len);
            const __bf16* dgrow = s_dg + (size_t)bl * G4P;
            const __bf16* wtr = sWT + (size_t)r * G4P;
            float acc = 0.f;
            for (int j8 = j0; j8 < j1; j8 += 8) {
              const bf16x8v dv = *reinterpret_cast<const bf16x8v*>(&dgrow[j8]);
              const bf16x8v wv = *reinterpret_cast<const bf16x8v*>(&wtr[j8]);
#pragma unroll
              for (int uu = 0; uu < 8; ++uu) acc += bf2f(dv[uu]) * bf2f(wv[uu]);
            }
            s_part[u] = acc;
          }
        }
        __syncthreads();
        if (rs > 0) {
          for (int i = tid; i < db * rs; i += kNT) {
            const int r = i % rs;
            const int bl = i / rs;
            float acc = 0.f;
#pragma unroll
            for (int seg = 0; seg < kSEG; ++seg) {
              acc += s_part[(bl * rs + r) * kSEG + seg];
            }
            const int b = b0 + bl;
            const float nd = notdone[(int64_t)t * B + b];
            const int64_t hi = (int64_t)b * H + r0 + r;
            const float v = nd * acc;
            scoped_store(&dst[hi], v);
            if (t == 0) dh_out[hi] = v;
          }
        }
        __syncthreads();
      }
    }
    agent_barrier(bar, bar + 1, nblocks, &lsense);
    buf ^= 1;
  }

  if (rs > 0) {
    for (int i = tid; i < B * rs; i += kNT) {
      const int b = i / rs, r = i % rs;
      dc_out[(int64_t)b * H + r0 + r] = dc_local[b * hs + r];
    }
  }
}

}  // namespace

// Host-side launch helpers (called from tbops.hip's lstm_unroll_{fwd,bwd}).

struct Lstm4Geometry {
  int hs;
  int nblocks;
  size_t fwd_lds;
  size_t bwd_lds;
};

Lstm4Geometry lstm4_geometry(int B, int H) {
  Lstm4Geometry g;
  // Measured sweep on MI355X (profiles/PROFILE_r2.md): ~173 workgroups is
  // the sweet spot for H=519 (more parallelism beats fewer barriers; the
  // kernel is latency- not bandwidth-bound). hs = slice rows per WG.
  g.hs = std::max(1, (H + 172) / 173);
  if (const char* e = std::getenv("TBAMD_LSTM_HS")) {
    const int v = std::atoi(e);
    if (v > 0) g.hs = v;  // experimental workgroup-count override
  }
  g.nblocks = (H + g.hs - 1) / g.hs;
  const int HP = (H + 7) & ~7;
  g.fwd_lds = (size_t)4 * g.hs * HP * 2 + (size_t)B * HP * 2 +
              (size_t)4 * g.hs * B * 4 + (size_t)B * g.hs * 4;
  const int G4P = (4 * H + 7) & ~7;
  g.bwd_lds = (size_t)g.hs * G4P * 2 + (size_t)kDB * G4P * 2 +
              (size_t)kDB * g.hs * kSEG * 4 + (size_t)B * g.hs * 4;
  return g;
}

void lstm4_fwd_launch(torch::Tensor precomp, torch::Tensor w_hh_bf,
                      torch::Tensor notdone, torch::Tensor h0,
                      torch::Tensor c0, torch::Tensor out,
                      torch::Tensor gates, torch::Tensor hm, torch::Tensor cm,
                      torch::Tensor c_stash, torch::Tensor hT,
                      torch::Tensor cT) {
  const int T = precomp.size(0), B = precomp.size(1);
  const int H = h0.size(1);
  auto g = lstm4_geometry(B, H);
  TORCH_CHECK(g.fwd_lds <= 160 * 1024, "lstm4 fwd LDS over budget (H=", H,
              " B=", B, ")");
  auto opts = precomp.options();
  auto stream = at::cuda::getCurrentCUDAStream();
  auto hglob = torch::empty({2, (int64_t)B * H}, opts);
  hglob[0].copy_(h0.reshape({-1}));
  auto bar = torch::zeros({2}, opts.dtype(torch::kInt32));

  const float* precomp_p = precomp.data_ptr<float>();
  const __bf16* w_p = reinterpret_cast<const __bf16*>(w_hh_bf.data_ptr());
  const float* nd_p = notdone.data_ptr<float>();
  const float* c0_p = c0.data_ptr<float>();
  int T_ = T, B_ = B, H_ = H, hs_ = g.hs, nb_ = g.nblocks;
  float* hglob_p = hglob.data_ptr<float>();
  int* bar_p = bar.data_ptr<int>();
  float* out_p = out.data_ptr<float>();
  float* gates_p = gates.data_ptr<float>();
  float* hm_p = hm.data_ptr<float>();
  float* cm_p = cm.data_ptr<float>();
  float* c_p = c_stash.data_ptr<float>();
  float* hT_p = hT.data_ptr<float>();
  float* cT_p = cT.data_ptr<float>();
  void* args[] = {&precomp_p, &w_p, &nd_p, &c0_p, &T_,    &B_,
                  &H_,        &hs_, &nb_,  &hglob_p, &bar_p, &out_p,
                  &gates_p,   &hm_p, &cm_p, &c_p,   &hT_p,  &cT_p};
  TORCH_CHECK(hipLaunchCooperativeKernel((const void*)lstm4_fwd_kernel,
                                         dim3(g.nblocks), dim3(kNT), args,
                                         g.fwd_lds, stream) == hipSuccess,
              "lstm4 fwd launch failed");
}

void lstm4_bwd_launch(torch::Tensor gates, torch::Tensor cm,
                      torch::Tensor c_stash, torch::Tensor w_hh_t_bf,
                      torch::Tensor notdone, torch::Tensor d_out,
                      torch::Tensor d_hT, torch::Tensor d_cT,
                      torch::Tensor dgates, torch::Tensor dh_out,
                      torch::Tensor dc_out) {
  const int T = gates.size(0), B = gates.size(1);
  const int H = d_hT.size(1);
  auto g = lstm4_geometry(B, H);
  TORCH_CHECK(g.bwd_lds <= 160 * 1024, "lstm4 bwd LDS over budget");
  auto opts = gates.options();
  auto stream = at::cuda::getCurrentCUDAStream();
  auto dhglob = torch::empty({2, (int64_t)B * H}, opts);
  dhglob[0].copy_(d_hT.reshape({-1}));
  auto dgates_comm = torch::empty({(int64_t)B * 4 * H}, opts);
  auto bar = torch::zeros({2}, opts.dtype(torch::kInt32));

  const float* gates_p = gates.data_ptr<float>();
  const float* cm_p = cm.data_ptr<float>();
  const float* c_p = c_stash.data_ptr<float>();
  const __bf16* wt_p = reinterpret_cast<const __bf16*>(w_hh_t_bf.data_ptr());
  const float* nd_p = notdone.data_ptr<float>();
  const float* dout_p = d_out.data_ptr<float>();
  const float* dcT_p = d_cT.data_ptr<float>();
  int T_ = T, B_ = B, H_ = H, hs_ = g.hs, nb_ = g.nblocks;
  float* dhglob_p = dhglob.data_ptr<float>();
  float* comm_p = dgates_comm.data_ptr<float>();
  int* bar_p = bar.data_ptr<int>();
  float* dg_p = dgates.data_ptr<float>();
  float* dh_p = dh_out.data_ptr<float>();
  float* dc_p = dc_out.data_ptr<float>();
  void* args[] = {&gates_p, &cm_p, &c_p,  &wt_p,     &nd_p,  &dout_p,
                  &dcT_p,   &T_,   &B_,   &H_,       &hs_,   &nb_,
                  &dhglob_p, &comm_p, &bar_p, &dg_p, &dh_p,  &dc_p};
  TORCH_CHECK(hipLaunchCooperativeKernel((const void*)lstm4_bwd_kernel,
                                         dim3(g.nblocks), dim3(kNT), args,
                                         g.bwd_lds, stream) == hipSuccess,
              "lstm4 bwd launch failed");
}

}  // namespace tbamd
