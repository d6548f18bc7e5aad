// ActorPool: one driver thread per environment, funneling single steps
// through the DynamicBatcher for batched GPU inference and complete
// [T+1, 1, ...] rollouts into the learner BatchingQueue.
//
// Capability parity with the reference ActorPool (ref:
// src/cc/actorpool.cc:342-564): rollouts overlap by one step (the last step
// of rollout k is the first of rollout k+1), and the agent state recorded
// with a rollout is the state *before* the inference of its first step.

#pragma once

#include <algorithm>
#include <atomic>
#include <condition_variable>
#include <cstring>
#include <future>
#include <mutex>
#include <memory>
#include <string>
#include <vector>

#include "env_transport.h"
#include "queues.h"

namespace tbruntime {

class ActorPool {
 public:
  ActorPool(int64_t unroll_length, std::shared_ptr<BatchingQueue> learner_queue,
            std::shared_ptr<DynamicBatcher> inference_batcher,
            std::vector<std::string> env_server_addresses,
            TensorNest initial_agent_state, int64_t seed_base = 0,
            bool use_obs_slab = false, int64_t rollout_budget_mb = 0,
            int64_t envs_per_thread = 1)
      : unroll_length_(unroll_length),
        learner_queue_(std::move(learner_queue)),
        inference_batcher_(std::move(inference_batcher)),
        addresses_(std::move(env_server_addresses)),
        initial_agent_state_(std::move(initial_agent_state)),
        seed_base_(seed_base),
        use_obs_slab_(use_obs_slab),
        envs_per_thread_(std::max<int64_t>(1, envs_per_thread)) {
    if (unroll_length_ < 1) {
      throw std::invalid_argument("unroll_length must be >= 1");
    }
    if (rollout_budget_mb > 0) {
      // Budget-sized pinned rollout ring with backpressure (replaces
      // unbounded ad-hoc pinned allocations); the learner queue recycles
      // slots after its H2D assembly completes. Floor at one slot per
      // actor + slack: an actor blocked on a full learner queue holds its
      // slot, so fewer slots than actors would throttle the whole pool
      // below the queue's own backpressure (measured: 7k wait wakeups per
      // bench run at 326 slots / 512 actors).
      rollout_pool_ = std::make_shared<PinnedSlabPool>(
          rollout_budget_mb * (1 << 20),
          /*min_slots=*/(int64_t)addresses_.size() + 64);
      learner_queue_->set_source_pool(rollout_pool_);
    }
  }

  // Pinned observation slab (slot per actor). Valid once the first env has
  // produced an observation; the inference runner gathers from it by slot
  // id instead of cat-ing per-request frame tensors.
  std::vector<torch::Tensor> obs_slab() {
    std::unique_lock<std::mutex> lk(slab_mu_);
    slab_cv_.wait(lk, [this] {
      return slab_ready_ || failed_ || !use_obs_slab_;
    });
    if (failed_ && !slab_ready_) {
      throw std::runtime_error("actor pool failed before producing obs");
    }
    if (!use_obs_slab_) return {};
    return {slab_frames_, slab_rew_, slab_done_};
  }

  // Blocks until every actor thread exits (via queue close or error).
  // Rethrows the first actor failure.
  void run() {
    std::vector<std::future<void>> futures;
    const size_t k = (size_t)envs_per_thread_;
    if (k > 1) {
      // Event-driven mode: each thread drives k env streams with
      // overlapped inference requests. 512 actor threads were measured
      // host-scheduling-bound (serve latency scaled with thread count,
      // profiles/PROFILE_r2.md); k envs per thread keeps the same env
      // parallelism with 1/k threads. Best for in-process synthetic envs
      // whose step() is microseconds; socket envs keep k == 1 so a slow
      // remote env cannot stall its neighbors.
      const size_t nthreads = (addresses_.size() + k - 1) / k;
      futures.reserve(nthreads);
      for (size_t t = 0; t < nthreads; ++t) {
        const size_t lo = t * k;
        const size_t hi = std::min(addresses_.size(), lo + k);
        futures.push_back(std::async(std::launch::async, [this, lo, hi] {
          try {
            loop_multi(lo, hi);
          } catch (...) {
            {
              std::lock_guard<std::mutex> lk(slab_mu_);
              failed_ = true;
            }
            slab_cv_.notify_all();
            throw;
          }
        }));
      }
      collect(futures);
      return;
    }
    futures.reserve(addresses_.size());
    for (size_t i = 0; i < addresses_.size(); ++i) {
      futures.push_back(std::async(std::launch::async, [this, i] {
        try {
          loop(addresses_[i], i);
        } catch (...) {
          // Wake obs_slab() waiters so setup can't hang on a dead actor.
          {
            std::lock_guard<std::mutex> lk(slab_mu_);
            failed_ = true;
          }
          slab_cv_.notify_all();
          throw;
        }
      }));
    }
    collect(futures);
  }

  uint64_t count() const { return step_count_.load(std::memory_order_relaxed); }

 private:
  // Copy this actor's observation into its pinned slot; the request nest
  // then carries only the slot id (gathered GPU-side by the runner).
  void fill_slot(int64_t slot, const TensorNest& env_outputs) {
    const auto& f = env_outputs.vector();
    const torch::Tensor frame = f[0].leaf();
    if (!slab_ready_) {
      std::lock_guard<std::mutex> lk(slab_mu_);
      if (!slab_ready_) {
        const int64_t n = (int64_t)addresses_.size();
        auto fshape = frame.sizes().vec();  // [1,1,C,H,W]
        std::vector<int64_t> slab_shape = {n};
        slab_shape.insert(slab_shape.end(), fshape.begin() + 2, fshape.end());
        // Pinned only when a GPU exists (pinned allocs require HIP);
        // CPU runs use the slab purely as shared host storage.
        auto hopts =
            torch::TensorOptions().pinned_memory(torch::cuda::is_available());
        slab_frames_ = torch::empty(slab_shape, hopts.dtype(torch::kUInt8));
        slab_rew_ = torch::zeros({n}, hopts.dtype(torch::kFloat32));
        slab_done_ = torch::zeros({n}, hopts.dtype(torch::kUInt8));
        slab_ready_ = true;
        slab_cv_.notify_all();
      }
    }
    const int64_t fsz = slab_frames_.stride(0);
    std::memcpy(
        static_cast<uint8_t*>(slab_frames_.data_ptr()) + slot * fsz,
        frame.data_ptr(), fsz);
    slab_rew_.data_ptr<float>()[slot] = f[1].leaf().item<float>();
    slab_done_.data_ptr<uint8_t>()[slot] = f[2].leaf().item<bool>() ? 1 : 0;
  }

  void collect(std::vector<std::future<void>>& futures) {
    std::exception_ptr first_error;
    for (auto& f : futures) {
      try {
        f.get();
      } catch (const ClosedQueue&) {
        // Normal shutdown path.
      } catch (...) {
        if (!first_error) first_error = std::current_exception();
      }
    }
    if (first_error) std::rethrow_exception(first_error);
  }

  // One thread, several env streams, overlapped inference futures. A
  // request is outstanding per env; completions are collected with a
  // short-timeout wait on one pending future plus a zero-timeout scan of
  // the rest (completions are correlated: batchmates finish together).
  void loop_multi(size_t lo, size_t hi) {
    struct EnvSlot {
      std::unique_ptr<EnvConnection> env;
      TensorNest env_outputs;
      TensorNest agent_state;
      TensorNest rollout_initial_state;
      TensorNest state_before;
      TensorNest slot_req;
      std::vector<TensorNest> rollout;
      std::future<TensorNest> pending;
      bool has_pending = false;
    };
    const size_t n = hi - lo;
    std::vector<EnvSlot> envs(n);
    for (size_t i = 0; i < n; ++i) {
      EnvSlot& e = envs[i];
      e.env = make_env_connection(addresses_[lo + i],
                                  seed_base_ + (int64_t)(lo + i) + 1);
      e.env_outputs = e.env->initial();
      e.agent_state = initial_agent_state_;
      e.rollout_initial_state = initial_agent_state_;
      e.rollout.reserve(unroll_length_ + 1);
      if (use_obs_slab_) {
        e.slot_req = TensorNest(torch::full(
            {1, 1}, (int64_t)(lo + i),
            torch::TensorOptions().dtype(torch::kInt32)));
      }
    }

    auto issue = [&](size_t i) {
      EnvSlot& e = envs[i];
      e.state_before = e.agent_state;
      TensorNest request;
      if (use_obs_slab_) {
        fill_slot((int64_t)(lo + i), e.env_outputs);
        request = TensorNest(TensorNest::vector_t{e.slot_req, e.agent_state});
      } else {
        request =
            TensorNest(TensorNest::vector_t{e.env_outputs, e.agent_state});
      }
      e.pending = inference_batcher_->compute_async(std::move(request));
      e.has_pending = true;
    };

    for (size_t i = 0; i < n; ++i) issue(i);

    for (;;) {
      // Block briefly on the first pending future, then sweep all.
      size_t first = n;
      for (size_t i = 0; i < n; ++i) {
        if (envs[i].has_pending) {
          first = i;
          break;
        }
      }
      if (first == n) throw std::logic_error("no pending inference");
      envs[first].pending.wait_for(std::chrono::microseconds(500));

      for (size_t i = 0; i < n; ++i) {
        EnvSlot& e = envs[i];
        if (!e.has_pending ||
            e.pending.wait_for(std::chrono::seconds(0)) !=
                std::future_status::ready) {
          continue;
        }
        e.has_pending = false;
        TensorNest result = e.pending.get();  // rethrows AsyncError
        if (!result.is_vector() || result.vector().size() != 2) {
          throw std::runtime_error(
              "inference must return ((action, ...), agent_state)");
        }
        TensorNest agent_outputs = result.vector()[0];
        e.agent_state = result.vector()[1];

        if (e.rollout.empty()) e.rollout_initial_state = e.state_before;
        e.rollout.push_back(
            TensorNest(TensorNest::vector_t{e.env_outputs, agent_outputs}));

        if (static_cast<int64_t>(e.rollout.size()) == unroll_length_ + 1) {
          flush_rollout(e.rollout, e.rollout_initial_state);
          TensorNest last = std::move(e.rollout.back());
          e.rollout.clear();
          e.rollout.push_back(std::move(last));
          e.rollout_initial_state = e.state_before;
        }

        const torch::Tensor& action = agent_outputs.front();
        e.env_outputs = e.env->step(action);
        step_count_.fetch_add(1, std::memory_order_relaxed);
        issue(i);
      }
    }
  }

  // Stack a complete [T+1] rollout into a pinned destination and enqueue.
  void flush_rollout(const std::vector<TensorNest>& rollout,
                     const TensorNest& rollout_initial_state) {
    std::vector<const TensorNest*> steps;
    steps.reserve(rollout.size());
    for (const auto& s : rollout) steps.push_back(&s);
    TensorNest stacked;
    if (rollout_pool_) {
      if (rollout_slot_bytes_ == 0) {
        int64_t total = 0;
        TensorNest::apply_columns(
            steps, [&total](const std::vector<torch::Tensor>& column) {
              int64_t rows = 0;
              for (const auto& t : column) rows += t.size(0);
              // Logical row bytes (exactly what carve() will allocate).
              // NOT stride(0): batch-split agent outputs are views whose
              // stride spans the whole serve batch, which varies per
              // serve — racing estimators would disagree on the total and
              // a late, larger store would overflow the fixed slot size.
              const int64_t row_numel =
                  column[0].numel() / std::max<int64_t>(1, column[0].size(0));
              const int64_t bytes =
                  rows * row_numel * column[0].element_size();
              total += (bytes + 255) & ~int64_t(255);
              return column[0];
            });
        rollout_slot_bytes_ = total + 4096;
      }
      auto slot = rollout_pool_->acquire(
          rollout_slot_bytes_, [this] { return learner_queue_->is_closed(); });
      stacked = TensorNest::apply_columns(
          steps, [&slot](const std::vector<torch::Tensor>& column) {
            auto shape = column[0].sizes().vec();
            int64_t rows = 0;
            for (const auto& t : column) rows += t.size(0);
            shape[0] = rows;
            torch::Tensor out = slot.carve(shape, column[0].scalar_type());
            torch::cat_out(out, column, 0);
            return out;
          });
    } else {
      stacked = TensorNest::apply_columns(
          steps, [](const std::vector<torch::Tensor>& column) {
            return cat_pinned(column, /*dim=*/0);
          });
    }
    learner_queue_->enqueue(TensorNest(
        TensorNest::vector_t{std::move(stacked), rollout_initial_state}));
  }

  void loop(const std::string& address, uint64_t seed) {
    auto env = make_env_connection(address, seed_base_ + seed + 1);

    TensorNest env_outputs = env->initial();
    TensorNest agent_state = initial_agent_state_;
    TensorNest rollout_initial_state = initial_agent_state_;

    // Slot-id request leaf (constant per actor, reused every step).
    TensorNest slot_req;
    if (use_obs_slab_) {
      slot_req = TensorNest(torch::full({1, 1}, (int64_t)seed,
                                        torch::TensorOptions().dtype(
                                            torch::kInt32)));
    }

    std::vector<TensorNest> rollout;
    rollout.reserve(unroll_length_ + 1);

    for (;;) {
      TensorNest state_before = agent_state;
      TensorNest request;
      if (use_obs_slab_) {
        fill_slot((int64_t)seed, env_outputs);
        request = TensorNest(TensorNest::vector_t{slot_req, agent_state});
      } else {
        request = TensorNest(TensorNest::vector_t{env_outputs, agent_state});
      }
      TensorNest result = inference_batcher_->compute(std::move(request));
      if (!result.is_vector() || result.vector().size() != 2) {
        throw std::runtime_error(
            "inference must return ((action, ...), agent_state)");
      }
      TensorNest agent_outputs = result.vector()[0];
      agent_state = result.vector()[1];

      if (rollout.empty()) rollout_initial_state = state_before;
      rollout.push_back(
          TensorNest(TensorNest::vector_t{env_outputs, agent_outputs}));

      if (static_cast<int64_t>(rollout.size()) == unroll_length_ + 1) {
        flush_rollout(rollout, rollout_initial_state);
        // Overlap by one step: the rollout we just sent ends with the step
        // whose pre-inference state is `state_before`.
        TensorNest last = std::move(rollout.back());
        rollout.clear();
        rollout.push_back(std::move(last));
        rollout_initial_state = state_before;
      }

      const torch::Tensor& action = agent_outputs.front();
      env_outputs = env->step(action);
      step_count_.fetch_add(1, std::memory_order_relaxed);
    }
  }

  const int64_t unroll_length_;
  std::shared_ptr<BatchingQueue> learner_queue_;
  std::shared_ptr<DynamicBatcher> inference_batcher_;
  std::vector<std::string> addresses_;
  TensorNest initial_agent_state_;
  const int64_t seed_base_;
  const bool use_obs_slab_;
  const int64_t envs_per_thread_;
  std::mutex slab_mu_;
  std::condition_variable slab_cv_;
  std::atomic<bool> slab_ready_{false};
  bool failed_ = false;
  torch::Tensor slab_frames_, slab_rew_, slab_done_;
  std::shared_ptr<PinnedSlabPool> rollout_pool_;
  std::atomic<int64_t> rollout_slot_bytes_{0};
  std::atomic<uint64_t> step_count_{0};
};

}  // namespace tbruntime
