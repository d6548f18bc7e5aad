// C++ inference engine: GIL-free behavior-model forwards.
//
// The reference serves inference from Python threads
// (ref: polybeast_learner.py:269-285); at MI355X throughput that path is
// GIL-bound (measured ~20-45 ms service latency for ~0.3 ms of GPU work).
// This runner consumes DynamicBatcher batches entirely in C++:
//   get_batch -> pinned-host cat -> H2D on a dedicated HIP stream ->
//   AtariNet forward (ATen ops; custom fused kernels swap in at the op
//   level) -> sampled action -> D2H -> set_outputs.
// Behavior-model weights are VIEWS of the learner's flat actor buffer, so
// weight sync stays one flat device copy with no runner involvement.

#pragma once

#include <ATen/cuda/CUDAContext.h>
#include <c10/cuda/CUDAGuard.h>
#include <torch/extension.h>

#include <atomic>
#include <chrono>
#include <cstdlib>
#include <mutex>
#include <set>
#include <thread>
#include <vector>

#include "../../ops/hip/atari_trunk.h"
#include "../../ops/hip/conv_mfma.h"
#include "queues.h"
#include "runtime_kernels.h"

namespace tbruntime {

class InferenceRunner {
 public:
  // weights order:
  //  shallow AtariNet (use_last_action=False):
  //   conv1.w, conv1.b, conv2.w, conv2.b, conv3.w, conv3.b,
  //   fc.w, fc.b, policy.w, policy.b, baseline.w, baseline.b
  //  deep IMPALA ResNet: per section (x3): conv.w, conv.b,
  //   res0.conv0.{w,b}, res0.conv1.{w,b}, res1.conv0.{w,b}, res1.conv1.{w,b}
  //   then fc.w, fc.b, policy.w, policy.b, baseline.w, baseline.b
  //  both: then per LSTM layer: w_ih, w_hh, b_ih, b_hh.
  InferenceRunner(std::shared_ptr<DynamicBatcher> batcher,
                  std::vector<torch::Tensor> weights, int64_t num_lstm_layers,
                  bool greedy = false, std::string model_type = "shallow")
      : batcher_(std::move(batcher)),
        weights_(std::move(weights)),
        num_lstm_layers_(num_lstm_layers),
        greedy_(greedy),
        deep_(model_type == "deep") {
    const size_t trunk = deep_ ? 30 : 6;
    head_base_ = trunk + 2;  // fc.w, fc.b come first after the trunk
    lstm_base_ = trunk + 6;  // fc(2) + policy(2) + baseline(2)
    TORCH_CHECK(weights_.size() >= lstm_base_ + 4 * (size_t)num_lstm_layers,
                "runner weight list too short for model/lstm config");
    TORCH_CHECK(weights_[0].is_cuda(), "runner weights must be on the GPU");
    device_ = weights_[0].device();
  }

  ~InferenceRunner() { stop(); }

  void start(int64_t num_threads) {
    running_ = true;
    for (int64_t i = 0; i < num_threads; ++i) {
      threads_.emplace_back([this] { loop(); });
    }
  }

  void stop() {
    running_ = false;
    if (batcher_ && !batcher_->is_closed()) {
      try {
        batcher_->close();
      } catch (...) {
      }
    }
    for (auto& t : threads_) {
      if (t.joinable()) t.join();
    }
    threads_.clear();
  }

  int64_t batches() const { return batches_.load(); }
  int64_t steps() const { return steps_.load(); }

  // Observation slab (from ActorPool::obs_slab) enabling the slot-id fast
  // path: requests carry only actor ids; frames/reward/done are gathered
  // GPU-side straight from the pinned slab.
  void set_obs_slab(torch::Tensor frames, torch::Tensor rew,
                    torch::Tensor done) {
    slab_frames_ = std::move(frames);
    slab_rew_ = std::move(rew);
    slab_done_ = std::move(done);
    frame_shape_ = std::vector<int64_t>(slab_frames_.sizes().begin() + 1,
                                        slab_frames_.sizes().end());
    has_slab_ = true;
  }

  // Learner weight sync happened: drop the cached bf16-packed trunk
  // weights (repacked lazily by the next serve).
  void mark_weights_dirty() { weights_version_.fetch_add(1); }

 private:
  void loop() {
    c10::cuda::CUDAGuard device_guard(device_);
    at::cuda::CUDAStream stream =
        at::cuda::getStreamFromPool(/*isHighPriority=*/true, device_.index());
    at::cuda::CUDAStreamGuard stream_guard(stream);
    torch::NoGradGuard no_grad;

    while (running_) {
      std::shared_ptr<DynamicBatcher::Batch> batch;
      try {
        batch = batcher_->get_batch();
      } catch (const ClosedQueue&) {
        return;
      }
      try {
        serve(*batch, stream);
      } catch (const std::exception& e) {
        // Drop this batch (its promises break with AsyncError and the
        // affected actors surface it) but keep the engine serving.
        fprintf(stderr, "InferenceRunner error (batch dropped): %s\n",
                e.what());
      }
    }
  }

  void trace(const char* what) {
    if (std::getenv("TBAMD_SERVE_TRACE")) {
      fprintf(stderr, "[serve %p] %s\n", (void*)this, what);
      fflush(stderr);
    }
  }

  void serve(DynamicBatcher::Batch& batch, at::cuda::CUDAStream& stream) {
    trace("begin");
    // Stage timers (enabled by TBAMD_SERVE_TIMINGS): cat / fwd / d2h+sync.
    const bool timing = serve_timing_;
    auto now_us = []() {
      return std::chrono::duration_cast<std::chrono::microseconds>(
                 std::chrono::steady_clock::now().time_since_epoch())
          .count();
    };
    int64_t t0 = timing ? now_us() : 0;
    TensorNest inputs = batch.get_inputs();
    trace("cat-done");
    int64_t t_cat = timing ? now_us() : 0;
    // Serialize the FIRST forward of each quantized batch size: concurrent
    // MIOpen solution-finds for a brand-new conv shape across streams have
    // been observed to fault; once found, the solution cache makes later
    // serves safe to run fully in parallel.
    const int64_t bq = (batch.size() + 63) / 64 * 64;
    // Deep 84x84 trunk serves on the hand-written MFMA 3x3 kernels
    // (conv_mfma.hip resnet_conv); MIOpen only for other geometries.
    const bool deep_mfma =
        deep_ && !aten_only_ && weights_[0].size(0) == 16 &&
        weights_[0].size(1) <= 8 && weights_[0].size(2) == 3 &&
        tbamd::resnet_conv_supported(16, 42, 16);
    std::unique_lock<std::mutex> warm_lock;
    if (deep_ && !deep_mfma) {
      // Only the MIOpen (deep ResNet) path needs first-serve-per-shape
      // serialization: concurrent solution-finds for a new conv shape
      // across streams have been observed to fault. The hand-written
      // trunk kernels have no such state.
      bool is_new;
      {
        std::lock_guard<std::mutex> g(warm_mu_);
        is_new = warmed_sizes_.count(bq) == 0;
      }
      if (is_new) {
        warm_lock = std::unique_lock<std::mutex>(serve_mu_);
      }
    }
    // inputs = ((frame, reward, done, ...), state)  [classic requests]
    //        = (slot_ids, state)                     [obs-slab requests]
    const auto& top = inputs.vector();
    const bool slot_mode = top[0].is_leaf();
    std::vector<torch::Tensor> state = top[1].flatten();  // [] or h,c [L,b,H]
    auto opts = torch::TensorOptions().device(device_);

    int64_t b, bp, C, H, W;
    torch::Tensor frames_p, rew, nd_gpu;
    if (slot_mode) {
      TORCH_CHECK(has_slab_, "slot-id request but no obs slab configured");
      const torch::Tensor ids = top[0].leaf();  // [1, b] int32
      b = ids.size(1);
      bp = (b + 63) / 64 * 64;
      auto g = gather_obs(slab_frames_, slab_rew_, slab_done_, ids, bp,
                          frame_shape_);
      frames_p = g[0];
      rew = g[1];
      nd_gpu = g[2];
      C = frame_shape_[0];
      H = frame_shape_[1];
      W = frame_shape_[2];
    } else {
      const auto& env = top[0].vector();
      const torch::Tensor frame = env[0].leaf();   // [1, b, C, H, W] u8
      const torch::Tensor reward = env[1].leaf();  // [1, b] f32
      const torch::Tensor done = env[2].leaf();    // [1, b] bool
      b = frame.size(1);
      // Quantize the compute batch to multiples of 64 so kernel/solution
      // caches see a handful of shapes, not every ragged batch size.
      bp = (b + 63) / 64 * 64;
      C = frame.size(2);
      H = frame.size(3);
      W = frame.size(4);
      frames_p = torch::empty({bp, C, H, W}, opts.dtype(torch::kUInt8));
      frames_p.narrow(0, 0, b).copy_(frame.reshape({b, C, H, W}),
                                     /*non_blocking=*/true);
      if (bp > b) frames_p.narrow(0, b, bp - b).zero_();
      rew = torch::empty({bp, 1}, opts.dtype(torch::kFloat32));
      rew.narrow(0, 0, b)
          .copy_(reward.to(opts.dtype(torch::kFloat32), true).reshape({b, 1}))
          .clamp_(-1, 1);
      if (bp > b) rew.narrow(0, b, bp - b).zero_();
      nd_gpu = (~done.reshape({b}))
                   .to(opts.dtype(torch::kFloat32), /*non_blocking=*/true);
    }

    torch::Tensor x;
    if (deep_ && deep_mfma && H == 84 && W == 84 && C <= 8) {
      // bf16 channels_last trunk, every conv on the MFMA template. Packed
      // bf16 weights cached until the learner's next sync.
      std::vector<torch::Tensor> wp;
      {
        std::lock_guard<std::mutex> g(pack_mu_);
        const int64_t v = weights_version_.load();
        if (deep_packed_version_ != v) {
          deep_packed_.clear();
          for (size_t wi = 0; wi < 30; wi += 2) {
            torch::Tensor w = weights_[wi];
            if (wi == 0 && w.size(1) < 8) {  // first conv: pad channels
              w = torch::cat(
                  {w, w.new_zeros({w.size(0), 8 - w.size(1), 3, 3})}, 1);
            }
            auto f = w.permute({0, 2, 3, 1}).reshape({w.size(0), -1})
                         .to(torch::kBFloat16);
            const int64_t k = f.size(1), kpad = (k + 31) / 32 * 32 - k;
            if (kpad) f = torch::cat({f, f.new_zeros({f.size(0), kpad})}, 1);
            deep_packed_.push_back(f.contiguous());
          }
          stream.synchronize();
          deep_packed_version_ = v;
        }
        wp = deep_packed_;
      }
      auto x8 = torch::empty({bp, 8, 84, 84}, opts.dtype(torch::kBFloat16),
                             torch::MemoryFormat::ChannelsLast);
      x8.zero_();
      x8.narrow(1, 0, C).copy_(frames_p, /*non_blocking=*/true);
      x8.mul_(1.0f / 255.0f);
      size_t wi = 0, pi = 0;
      auto conv = [&](torch::Tensor t, int64_t ci, int64_t hw, int64_t co) {
        auto o = tbamd::resnet_conv(t, wp[pi], weights_[wi + 1], ci, hw, co,
                                    /*fwd=*/true);
        ++pi;
        wi += 2;
        return o.permute({0, 3, 1, 2});
      };
      const int64_t widths[3] = {16, 32, 32};
      int64_t hw = 84, ci = 8;
      torch::Tensor t = x8;
      for (int sec = 0; sec < 3; ++sec) {
        t = conv(t, ci, hw, widths[sec]);
        t = at::max_pool2d(t, 3, 2, 1);
        hw = (hw + 1) / 2;
        ci = widths[sec];
        for (int res = 0; res < 2; ++res) {
          torch::Tensor y = conv(at::relu(t), ci, hw, ci);
          y = conv(at::relu(y), ci, hw, ci);
          t = t + y;
        }
      }
      x = at::relu(t).to(torch::kFloat32).reshape({bp, -1});
    } else if (deep_) {
      x = frames_p.to(torch::kFloat32).mul_(1.0f / 255.0f);
      size_t wi = 0;
      for (int sec = 0; sec < 3; ++sec) {
        x = at::conv2d(x, weights_[wi], weights_[wi + 1], 1, 1);
        wi += 2;
        x = at::max_pool2d(x, 3, 2, 1);
        for (int res = 0; res < 2; ++res) {
          torch::Tensor y = at::conv2d(at::relu(x), weights_[wi],
                                       weights_[wi + 1], 1, 1);
          y = at::conv2d(y.relu_(), weights_[wi + 2], weights_[wi + 3], 1, 1);
          wi += 4;
          x = x + y;
        }
      }
      x = at::relu(x).reshape({bp, -1});
    } else if (!aten_only_ && ((C == 4 && H == 84 && W == 84) ||
                               (C == 3 && H == 210 && W == 160))) {
      // MFMA implicit-GEMM trunk (bf16 operands, fp32 accumulate) — flat
      // per-batch cost at every dynamic batch size, unlike the per-sample
      // fused kernel whose grid is the batch (underfills the 256 CUs below
      // ~512 samples and was measured dominating GPU time at small
      // batches: profiles/PROFILE_r2.md).
      torch::Tensor w1p, w2p, w3p;
      {
        // bf16-packed trunk weights, cached until the learner's next sync
        // (mark_weights_dirty). The packing thread synchronizes its stream
        // before publishing so other serve streams read complete data.
        std::lock_guard<std::mutex> g(pack_mu_);
        const int64_t v = weights_version_.load();
        if (packed_version_ != v) {
          w1p_ = weights_[0].reshape({32, -1}).to(torch::kBFloat16)
                     .contiguous();
          w2p_ = weights_[2].permute({0, 2, 3, 1}).reshape({64, -1})
                     .to(torch::kBFloat16).contiguous();
          w3p_ = weights_[4].permute({0, 2, 3, 1}).reshape({64, -1})
                     .to(torch::kBFloat16).contiguous();
          stream.synchronize();
          packed_version_ = v;
        }
        w1p = w1p_;
        w2p = w2p_;
        w3p = w3p_;
      }
      trace("trunk");
      x = tbamd::conv_trunk_fwd(frames_p, w1p, weights_[1], w2p, weights_[3],
                                w3p, weights_[5], /*want_stash=*/false)[0];
      trace("trunk-done");
    } else if (!aten_only_ && bp <= 384 &&
               tbamd::atari_trunk_supported(C, H, W)) {
      // Hand-written fused CDNA4 conv trunk: one kernel for the u8
      // normalize + 3 convs (non-84x84 geometries).
      x = tbamd::atari_trunk_fwd(frames_p, weights_[0], weights_[1],
                                 weights_[2], weights_[3], weights_[4],
                                 weights_[5], /*save_for_backward=*/false)[0];
    } else {
      x = frames_p.to(torch::kFloat32).mul_(1.0f / 255.0f);
      x = at::conv2d(x, weights_[0], weights_[1], /*stride=*/4).relu_();
      x = at::conv2d(x, weights_[2], weights_[3], 2).relu_();
      x = at::conv2d(x, weights_[4], weights_[5], 1).relu_();
      x = x.reshape({bp, -1});
    }
    x = at::linear(x, weights_[head_base_ - 2], weights_[head_base_ - 1])
            .relu_();

    if (!aten_only_ && num_lstm_layers_ == 0 &&
        weights_[head_base_].size(0) <= 64 &&
        weights_[head_base_].size(1) == x.size(1) + 1) {
      // Fused heads + Gumbel sample, written straight into pinned host
      // buffers: replaces cat/2x linear/rand/log/argmax/3x D2H.
      const int64_t seed =
          greedy_ ? 0 : (int64_t)(seed_ctr_.fetch_add(1) * 0x9E3779B97F4A7C15ull);
      trace("heads");
      auto hs = fused_heads_sample(
          x, rew, weights_[head_base_], weights_[head_base_ + 1],
          weights_[head_base_ + 2], weights_[head_base_ + 3], b, greedy_,
          seed);
      trace("heads-sync");
      stream.synchronize();
      trace("heads-done");
      int64_t t_fwd2 = timing ? now_us() : 0;
      TensorNest::vector_t agent_out{
          TensorNest(hs[0].reshape({1, b})),
          TensorNest(hs[1].reshape({1, b, -1})),
          TensorNest(hs[2].reshape({1, b})),
      };
      batch.set_outputs(TensorNest(TensorNest::vector_t{
          TensorNest(std::move(agent_out)), TensorNest(TensorNest::vector_t{})}));
      batches_.fetch_add(1, std::memory_order_relaxed);
      steps_.fetch_add(b, std::memory_order_relaxed);
      if (timing) {
        t_cat_us_.fetch_add(t_cat - t0, std::memory_order_relaxed);
        t_fwd_us_.fetch_add(t_fwd2 - t_cat, std::memory_order_relaxed);
        const int64_t n = timed_batches_.fetch_add(1) + 1;
        if (n % 500 == 0) {
          fprintf(stderr,
                  "[serve timings over %lld batches] cat=%.0fus "
                  "fwd+sync=%.0fus\n",
                  (long long)n, (double)t_cat_us_.load() / n,
                  (double)t_fwd_us_.load() / n);
        }
      }
      if (warm_lock.owns_lock()) {
        std::lock_guard<std::mutex> g(warm_mu_);
        warmed_sizes_.insert(bq);
      }
      return;
    }

    torch::Tensor core = at::cat({x, rew}, 1);

    std::vector<torch::Tensor> new_state_gpu;
    if (num_lstm_layers_ > 0) {
      TORCH_CHECK(state.size() == 2, "lstm runner needs (h, c) state");
      const int64_t L = state[0].size(0);
      const int64_t H = state[0].size(2);
      torch::Tensor nd = nd_gpu.narrow(0, 0, b).reshape({1, b, 1});
      auto pad_state = [&](const torch::Tensor& s) {
        torch::Tensor g = torch::zeros({L, bp, H}, opts.dtype(torch::kFloat32));
        g.narrow(1, 0, b).copy_(
            s.to(opts.dtype(torch::kFloat32), true) * nd);
        return g;
      };
      torch::Tensor h = pad_state(state[0]);
      torch::Tensor c = pad_state(state[1]);
      torch::Tensor layer_in = core;
      std::vector<torch::Tensor> hs, cs;
      for (int64_t l = 0; l < num_lstm_layers_; ++l) {
        const auto& w_ih = weights_[lstm_base_ + 4 * l];
        const auto& w_hh = weights_[lstm_base_ + 4 * l + 1];
        const auto& b_ih = weights_[lstm_base_ + 4 * l + 2];
        const auto& b_hh = weights_[lstm_base_ + 4 * l + 3];
        torch::Tensor gates = at::addmm(b_ih + b_hh, layer_in, w_ih.t())
                                  .addmm_(h[l], w_hh.t());
        auto chunks = gates.chunk(4, 1);
        torch::Tensor ig = at::sigmoid(chunks[0]);
        torch::Tensor fg = at::sigmoid(chunks[1]);
        torch::Tensor gg = at::tanh(chunks[2]);
        torch::Tensor og = at::sigmoid(chunks[3]);
        torch::Tensor c_new = fg * c[l] + ig * gg;
        torch::Tensor h_new = og * at::tanh(c_new);
        hs.push_back(h_new);
        cs.push_back(c_new);
        layer_in = h_new;
      }
      core = layer_in;
      new_state_gpu = {at::stack(hs).narrow(1, 0, b),
                       at::stack(cs).narrow(1, 0, b)};
    }

    torch::Tensor logits =
        at::linear(core, weights_[head_base_], weights_[head_base_ + 1])
            .narrow(0, 0, b);
    torch::Tensor baseline =
        at::linear(core, weights_[head_base_ + 2], weights_[head_base_ + 3])
            .narrow(0, 0, b)
            .reshape({b});
    torch::Tensor action;
    if (greedy_) {
      action = at::argmax(logits, -1);
    } else {
      // Gumbel-argmax categorical sampling: exactly softmax(logits), no
      // host sync, three elementwise kernels (at::multinomial stalls).
      torch::Tensor u = at::rand_like(logits).clamp_(1e-20, 1.0);
      action = at::argmax(logits - at::log(-at::log(u)), -1);
    }

    // D2H into pinned host tensors, one stream sync for the whole batch.
    auto to_host = [&](const torch::Tensor& t) {
      torch::Tensor out = torch::empty(
          t.sizes(), t.options().device(torch::kCPU).pinned_memory(
                         pinned_memory_wanted()));
      out.copy_(t, /*non_blocking=*/true);
      return out;
    };
    int64_t t_fwd = timing ? now_us() : 0;
    torch::Tensor a_h = to_host(action);
    torch::Tensor l_h = to_host(logits);
    torch::Tensor b_h = to_host(baseline);
    std::vector<torch::Tensor> state_h;
    for (const auto& s : new_state_gpu) state_h.push_back(to_host(s));
    stream.synchronize();
    if (timing) {
      const int64_t t_sync = now_us();
      t_cat_us_.fetch_add(t_cat - t0, std::memory_order_relaxed);
      t_fwd_us_.fetch_add(t_fwd - t_cat, std::memory_order_relaxed);
      t_d2h_us_.fetch_add(t_sync - t_fwd, std::memory_order_relaxed);
      const int64_t n = timed_batches_.fetch_add(1) + 1;
      if (n % 500 == 0) {
        fprintf(stderr,
                "[serve timings over %lld batches] cat=%.0fus fwd=%.0fus "
                "d2h+sync=%.0fus\n",
                (long long)n, (double)t_cat_us_.load() / n,
                (double)t_fwd_us_.load() / n, (double)t_d2h_us_.load() / n);
      }
    }

    TensorNest::vector_t agent_out{
        TensorNest(a_h.reshape({1, b})),
        TensorNest(l_h.reshape({1, b, -1})),
        TensorNest(b_h.reshape({1, b})),
    };
    TensorNest::vector_t state_out;
    for (const auto& s : state_h) state_out.emplace_back(s);
    batch.set_outputs(TensorNest(TensorNest::vector_t{
        TensorNest(std::move(agent_out)), TensorNest(std::move(state_out))}));

    batches_.fetch_add(1, std::memory_order_relaxed);
    steps_.fetch_add(b, std::memory_order_relaxed);
    if (warm_lock.owns_lock()) {
      std::lock_guard<std::mutex> g(warm_mu_);
      warmed_sizes_.insert(bq);
    }
  }

  bool has_slab_ = false;
  torch::Tensor slab_frames_, slab_rew_, slab_done_;
  std::vector<int64_t> frame_shape_;
  std::mutex pack_mu_;
  torch::Tensor w1p_, w2p_, w3p_;
  std::vector<torch::Tensor> deep_packed_;
  int64_t deep_packed_version_ = -1;
  int64_t packed_version_ = -1;
  std::atomic<int64_t> weights_version_{0};
  std::atomic<uint64_t> seed_ctr_{1};
  // Benchmarking: serve everything through library ops (the strategy
  // comparison in scripts/inference_speed_profiling.py).
  const bool aten_only_ = std::getenv("TBAMD_RUNNER_ATEN") != nullptr;
  const bool serve_timing_ = std::getenv("TBAMD_SERVE_TIMINGS") != nullptr;
  std::atomic<int64_t> t_cat_us_{0}, t_fwd_us_{0}, t_d2h_us_{0};
  std::atomic<int64_t> timed_batches_{0};
  std::mutex warm_mu_;
  std::mutex serve_mu_;
  std::set<int64_t> warmed_sizes_;
  std::shared_ptr<DynamicBatcher> batcher_;
  std::vector<torch::Tensor> weights_;
  const int64_t num_lstm_layers_;
  const bool greedy_;
  bool deep_ = false;
  size_t head_base_ = 8;
  size_t lstm_base_ = 12;
  torch::Device device_ = torch::kCPU;
  std::atomic<bool> running_{false};
  std::atomic<int64_t> batches_{0};
  std::atomic<int64_t> steps_{0};
  std::vector<std::thread> threads_;
};

}  // namespace tbruntime
