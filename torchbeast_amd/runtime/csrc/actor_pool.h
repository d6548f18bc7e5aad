// ActorPool: one driver thread per environment, funneling single steps
// through the DynamicBatcher for batched GPU inference and complete
// [T+1, 1, ...] rollouts into the learner BatchingQueue.
//
// Capability parity with the reference ActorPool (ref:
// src/cc/actorpool.cc:342-564): rollouts overlap by one step (the last step
// of rollout k is the first of rollout k+1), and the agent state recorded
// with a rollout is the state *before* the inference of its first step.

#pragma once

#include <atomic>
#include <future>
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
            TensorNest initial_agent_state, int64_t seed_base = 0)
      : unroll_length_(unroll_length),
        learner_queue_(std::move(learner_queue)),
        inference_batcher_(std::move(inference_batcher)),
        addresses_(std::move(env_server_addresses)),
        initial_agent_state_(std::move(initial_agent_state)),
        seed_base_(seed_base) {
    if (unroll_length_ < 1) {
      throw std::invalid_argument("unroll_length must be >= 1");
    }
  }

  // Blocks until every actor thread exits (via queue close or error).
  // Rethrows the first actor failure.
  void run() {
    std::vector<std::future<void>> futures;
    futures.reserve(addresses_.size());
    for (size_t i = 0; i < addresses_.size(); ++i) {
      futures.push_back(std::async(
          std::launch::async, [this, i] { loop(addresses_[i], i); }));
    }
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

  uint64_t count() const { return step_count_.load(std::memory_order_relaxed); }

 private:
  void loop(const std::string& address, uint64_t seed) {
    auto env = make_env_connection(address, seed_base_ + seed + 1);

    TensorNest env_outputs = env->initial();
    TensorNest agent_state = initial_agent_state_;
    TensorNest rollout_initial_state = initial_agent_state_;

    std::vector<TensorNest> rollout;
    rollout.reserve(unroll_length_ + 1);

    for (;;) {
      TensorNest state_before = agent_state;
      TensorNest result = inference_batcher_->compute(
          TensorNest(TensorNest::vector_t{env_outputs, agent_state}));
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
        std::vector<const TensorNest*> steps;
        steps.reserve(rollout.size());
        for (const auto& s : rollout) steps.push_back(&s);
        // Pinned slab so the learner-side dequeue can DMA it to the GPU.
        TensorNest stacked = TensorNest::apply_columns(
            steps, [](const std::vector<torch::Tensor>& column) {
              return cat_pinned(column, /*dim=*/0);
            });
        learner_queue_->enqueue(TensorNest(
            TensorNest::vector_t{std::move(stacked), rollout_initial_state}));

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
  std::atomic<uint64_t> step_count_{0};
};

}  // namespace tbruntime
