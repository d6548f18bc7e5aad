// Batching queues for the actor-learner runtime.
//
// Capability parity with the reference's BatchingQueue / DynamicBatcher
// (ref: src/cc/actorpool.cc:72-340), redesigned for MI355X:
// - dequeue/batch assembly concatenates into HIP-*pinned* host tensors
//   (cat_out into a pinned destination) so the learner/inference H2D copy
//   is a true DMA on a side stream, instead of pageable torch::cat output.
// - one generic bounded MPMC queue underlies both the learner rollout queue
//   and the inference request queue.

#pragma once

#include <ATen/cuda/CUDAContext.h>
#include <ATen/cuda/CUDAEvent.h>
#include <torch/extension.h>

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <deque>
#include <functional>
#include <future>
#include <map>
#include <memory>
#include <mutex>
#include <optional>
#include <stdexcept>
#include <vector>

#include "nest.h"

namespace tbruntime {

using TensorNest = Nest<torch::Tensor>;

inline int64_t now_us() {
  return std::chrono::duration_cast<std::chrono::microseconds>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

// Lightweight perf counters (atomics; read via .stats() from Python).
struct StageStats {
  std::atomic<int64_t> n{0};
  std::atomic<int64_t> sum_us{0};
  void add(int64_t us) {
    n.fetch_add(1, std::memory_order_relaxed);
    sum_us.fetch_add(us, std::memory_order_relaxed);
  }
  void reset() {
    n.store(0, std::memory_order_relaxed);
    sum_us.store(0, std::memory_order_relaxed);
  }
};

class ClosedQueue : public std::runtime_error {
 public:
  using std::runtime_error::runtime_error;
};

class AsyncError : public std::runtime_error {
 public:
  using std::runtime_error::runtime_error;
};

inline bool pinned_memory_wanted() {
  static const bool wanted = [] {
    if (std::getenv("TBAMD_NO_PIN")) return false;
    return torch::cuda::is_available();
  }();
  return wanted;
}

// Concatenate leaves along `dim` into a pinned destination when a GPU is
// present (so the later .to(device, non_blocking=True) is DMA).
inline torch::Tensor cat_pinned(const std::vector<torch::Tensor>& tensors,
                                int64_t dim) {
  TORCH_CHECK(!tensors.empty(), "cat_pinned: empty tensor list");
  if (!pinned_memory_wanted() || tensors[0].is_cuda()) {
    return torch::cat(tensors, dim);
  }
  auto shape = tensors[0].sizes().vec();
  int64_t total = 0;
  for (const auto& t : tensors) total += t.size(dim);
  shape[dim] = total;
  int64_t bytes = tensors[0].element_size();
  for (auto d : shape) bytes *= d;
  if (bytes < 65536) {
    // Small outputs (slot ids, scalars): ragged pinned allocations would
    // hit hipHostMalloc (a device-wide sync) for every new size; a plain
    // cat + pageable H2D is far cheaper at this size.
    return torch::cat(tensors, dim);
  }
  torch::Tensor out = torch::empty(
      shape, tensors[0].options().pinned_memory(true));
  torch::cat_out(out, tensors, dim);
  return out;
}

// ---------------------------------------------------------------------------
// PinnedSlabPool: fixed-capacity pinned rollout buffering with backpressure.
//
// One pinned slab carved into equal rollout-sized slots (slot size fixed by
// the first acquire; every rollout of a run has the same shape). An actor
// acquires a slot, cats its finished rollout into slab VIEWS (so
// .is_pinned() stays true and H2D copies stay DMA), and enqueues them.
// Rollout tensors never escape the learner-side BatchingQueue (both
// assembly modes produce fresh batched outputs), so slot recycling is
// driven by the queue alone: after assembling a batch it calls
// mark_consumed(ptr, event) per source leaf, and the slot returns to the
// free list once that dequeue's copy event has completed (immediately in
// CPU mode). acquire() blocks when the budget is exhausted — this is the
// backpressure that bounds rollout memory to the configured budget
// (BASELINE.json north star: buffering sized against a memory budget, not
// unbounded ad-hoc allocations).
// ---------------------------------------------------------------------------

class PinnedSlabPool {
 public:
  explicit PinnedSlabPool(int64_t budget_bytes, int64_t min_slots = 64)
      : budget_bytes_(budget_bytes), min_slots_(min_slots) {}

  // Handle for filling one slot; carve() returns pinned slab views.
  struct Slot {
    PinnedSlabPool* pool;
    int64_t index;
    int64_t used = 0;

    torch::Tensor carve(std::vector<int64_t> shape, torch::ScalarType dtype) {
      int64_t numel = 1;
      for (auto d : shape) numel *= d;
      const int64_t nbytes = numel * (int64_t)torch::elementSize(dtype);
      const int64_t aligned = (used + 255) & ~int64_t(255);
      TORCH_CHECK(aligned + nbytes <= pool->slot_bytes_,
                  "rollout slot overflow");
      torch::Tensor flat = pool->slab_.narrow(
          0, index * pool->slot_bytes_ + aligned, nbytes);
      used = aligned + nbytes;
      return flat.view(dtype).reshape(shape);
    }
  };

  // Blocks until a slot is free; `cancelled` (checked every 50 ms) lets
  // shutdown unwind the actor threads.
  Slot acquire(int64_t slot_bytes, const std::function<bool()>& cancelled) {
    std::unique_lock<std::mutex> lk(mu_);
    if (!slab_.defined()) {
      slot_bytes_ = (slot_bytes + 4095) & ~int64_t(4095);
      int64_t slots = budget_bytes_ / slot_bytes_;
      if (slots < min_slots_) slots = min_slots_;
      slab_ = torch::empty(
          {slots * slot_bytes_},
          torch::TensorOptions()
              .dtype(torch::kUInt8)
              .pinned_memory(pinned_memory_wanted()));
      for (int64_t i = slots - 1; i >= 0; --i) free_.push_back(i);
      total_slots_ = slots;
    }
    TORCH_CHECK(slot_bytes <= slot_bytes_,
                "rollout larger than the slab slot size");
    for (;;) {
      reap_pending();
      if (!free_.empty()) {
        const int64_t idx = free_.back();
        free_.pop_back();
        return Slot{this, idx, 0};
      }
      backpressure_waits_.fetch_add(1, std::memory_order_relaxed);
      cv_.wait_for(lk, std::chrono::milliseconds(50));
      if (cancelled && cancelled()) {
        throw ClosedQueue("slab pool acquire cancelled");
      }
    }
  }

  // Consumer-side: `ptr` was read by copies whose completion `ev` tracks
  // (ev == nullptr for synchronous CPU assembly). Idempotent per slot per
  // dequeue: all leaves of a rollout carry the same event.
  void mark_consumed(const void* ptr,
                     const std::shared_ptr<at::cuda::CUDAEvent>& ev) {
    std::lock_guard<std::mutex> lk(mu_);
    if (!slab_.defined()) return;
    const uint8_t* base = slab_.data_ptr<uint8_t>();
    const auto* p = static_cast<const uint8_t*>(ptr);
    if (p < base || p >= base + total_slots_ * slot_bytes_) return;
    const int64_t idx = (p - base) / slot_bytes_;
    if (ev == nullptr) {
      if (!in_list(free_, idx)) free_.push_back(idx);
    } else if (!in_pending(idx)) {
      pending_.emplace_back(idx, ev);
    }
    cv_.notify_all();
  }

  bool manages(const void* ptr) const {
    std::lock_guard<std::mutex> lk(mu_);
    if (!slab_.defined()) return false;
    const uint8_t* base = slab_.data_ptr<uint8_t>();
    const auto* p = static_cast<const uint8_t*>(ptr);
    return p >= base && p < base + total_slots_ * slot_bytes_;
  }

  std::map<std::string, double> stats() const {
    std::lock_guard<std::mutex> lk(mu_);
    std::map<std::string, double> out;
    out["slab_slots"] = (double)total_slots_;
    out["slab_slot_bytes"] = (double)slot_bytes_;
    out["slab_free"] = (double)free_.size();
    out["slab_pending"] = (double)pending_.size();
    out["slab_backpressure_waits"] = (double)backpressure_waits_.load();
    return out;
  }

 private:
  static bool in_list(const std::vector<int64_t>& v, int64_t x) {
    for (auto e : v) {
      if (e == x) return true;
    }
    return false;
  }
  bool in_pending(int64_t idx) const {
    for (const auto& p : pending_) {
      if (p.first == idx) return true;
    }
    return false;
  }
  void reap_pending() {
    for (auto it = pending_.begin(); it != pending_.end();) {
      if (it->second->query()) {
        free_.push_back(it->first);
        it = pending_.erase(it);
      } else {
        ++it;
      }
    }
  }

  const int64_t budget_bytes_;
  const int64_t min_slots_;
  mutable std::mutex mu_;
  std::condition_variable cv_;
  torch::Tensor slab_;
  int64_t slot_bytes_ = 0;
  int64_t total_slots_ = 0;
  std::vector<int64_t> free_;
  std::vector<std::pair<int64_t, std::shared_ptr<at::cuda::CUDAEvent>>>
      pending_;
  std::atomic<int64_t> backpressure_waits_{0};
};

inline TensorNest batch_nests(const std::vector<const TensorNest*>& nests,
                              int64_t batch_dim) {
  return TensorNest::apply_columns(
      nests, [batch_dim](const std::vector<torch::Tensor>& column) {
        return cat_pinned(column, batch_dim);
      });
}

// Batch assembly with the destination ON THE GPU: each (pinned) source
// tensor is copied with one async H2D into its slice of a device-resident
// output. Replaces {CPU cat -> one huge H2D} with B small DMAs that
// interleave with inference traffic, and eliminates the host-side memcpy
// entirely (the "rollouts DMA straight into the learner" design,
// SURVEY.md §2.2).
inline TensorNest batch_nests_to_device(
    const std::vector<const TensorNest*>& nests, int64_t batch_dim,
    torch::Device device) {
  return TensorNest::apply_columns(
      nests,
      [batch_dim, device](const std::vector<torch::Tensor>& column) {
        auto shape = column[0].sizes().vec();
        int64_t total = 0;
        for (const auto& t : column) total += t.size(batch_dim);
        shape[batch_dim] = total;
        torch::Tensor out =
            torch::empty(shape, column[0].options().device(device));
        int64_t offset = 0;
        for (const auto& t : column) {
          out.narrow(batch_dim, offset, t.size(batch_dim))
              .copy_(t, /*non_blocking=*/true);
          offset += t.size(batch_dim);
        }
        return out;
      });
}

// ---------------------------------------------------------------------------
// Generic bounded MPMC queue.
// ---------------------------------------------------------------------------

template <typename Item>
class BoundedQueue {
 public:
  explicit BoundedQueue(std::optional<int64_t> max_size = std::nullopt)
      : max_size_(max_size) {}

  void enqueue(Item item) {
    {
      std::unique_lock<std::mutex> lock(mu_);
      not_full_.wait(lock, [this] {
        return closed_ || !max_size_ ||
               static_cast<int64_t>(items_.size()) < *max_size_;
      });
      if (closed_) throw ClosedQueue("enqueue to closed queue");
      items_.push_back(std::move(item));
    }
    not_empty_.notify_one();
  }

  // Block until at least min_n items (or timeout with >=1, or close).
  // Returns up to max_n items. Throws ClosedQueue when closed and drained.
  std::vector<Item> dequeue_many(int64_t min_n, int64_t max_n,
                                 std::optional<std::chrono::milliseconds>
                                     timeout = std::nullopt) {
    std::unique_lock<std::mutex> lock(mu_);
    auto have_min = [this, min_n] {
      return closed_ || static_cast<int64_t>(items_.size()) >= min_n;
    };
    if (timeout) {
      // After the deadline, settle for any non-empty prefix.
      if (!not_empty_.wait_for(lock, *timeout, have_min)) {
        not_empty_.wait(lock, [this] { return closed_ || !items_.empty(); });
      }
    } else {
      not_empty_.wait(lock, have_min);
    }
    if (items_.empty()) {
      // Only reachable when closed.
      throw ClosedQueue("queue is closed");
    }
    int64_t n = std::min<int64_t>(items_.size(), max_n);
    std::vector<Item> out;
    out.reserve(n);
    for (int64_t i = 0; i < n; ++i) {
      out.push_back(std::move(items_.front()));
      items_.pop_front();
    }
    lock.unlock();
    not_full_.notify_all();
    return out;
  }

  int64_t size() const {
    std::lock_guard<std::mutex> lock(mu_);
    return items_.size();
  }

  void close() {
    {
      std::lock_guard<std::mutex> lock(mu_);
      if (closed_) throw ClosedQueue("queue was already closed");
      closed_ = true;
    }
    not_empty_.notify_all();
    not_full_.notify_all();
  }

  bool is_closed() const {
    std::lock_guard<std::mutex> lock(mu_);
    return closed_;
  }

 private:
  mutable std::mutex mu_;
  std::condition_variable not_empty_;
  std::condition_variable not_full_;
  std::deque<Item> items_;
  std::optional<int64_t> max_size_;
  bool closed_ = false;
};

// ---------------------------------------------------------------------------
// Learner-facing queue of rollouts, batched along batch_dim on dequeue.
// ---------------------------------------------------------------------------

class BatchingQueue {
 public:
  BatchingQueue(int64_t batch_dim = 0,
                std::optional<int64_t> minimum_batch_size = std::nullopt,
                std::optional<int64_t> maximum_batch_size = std::nullopt,
                std::optional<int64_t> timeout_ms = std::nullopt,
                bool check_inputs = true,
                std::optional<int64_t> maximum_queue_size = std::nullopt,
                std::optional<std::string> output_device = std::nullopt)
      : output_device_(output_device
                           ? std::optional<torch::Device>(
                                 torch::Device(*output_device))
                           : std::nullopt),
        batch_dim_(batch_dim),
        min_batch_size_(minimum_batch_size ? *minimum_batch_size : 1),
        max_batch_size_(maximum_batch_size
                            ? *maximum_batch_size
                            : std::numeric_limits<int64_t>::max()),
        timeout_(timeout_ms
                     ? std::optional<std::chrono::milliseconds>(
                           std::chrono::milliseconds(*timeout_ms))
                     : std::nullopt),
        check_inputs_(check_inputs),
        queue_(maximum_queue_size) {
    if (min_batch_size_ < 1) {
      throw std::invalid_argument("Min batch size must be >= 1");
    }
    if (max_batch_size_ < min_batch_size_) {
      throw std::invalid_argument(
          "Max batch size must be >= min batch size");
    }
    if (maximum_queue_size && *maximum_queue_size < 1) {
      throw std::invalid_argument("Max queue size must be >= 1");
    }
  }

  int64_t batch_dim() const { return batch_dim_; }

  void enqueue(TensorNest item) {
    if (check_inputs_) {
      if (item.empty()) throw std::invalid_argument("Empty input");
      item.for_each([this](const torch::Tensor& t) {
        if (t.dim() <= batch_dim_) {
          throw std::invalid_argument(
              "Enqueued tensors must have more than batch_dim dims");
        }
      });
    }
    queue_.enqueue(std::move(item));
  }

  // One batched nest (cat along batch_dim) + the number of rollouts in it.
  std::pair<TensorNest, int64_t> dequeue_many() {
    const int64_t t0 = now_us();
    std::vector<TensorNest> items =
        queue_.dequeue_many(min_batch_size_, max_batch_size_, timeout_);
    const int64_t t1 = now_us();
    std::vector<const TensorNest*> ptrs;
    ptrs.reserve(items.size());
    for (const auto& n : items) ptrs.push_back(&n);
    TensorNest batched;
    if (output_device_ && output_device_->is_cuda()) {
      // Assemble directly on the GPU: async copies on a dedicated copy
      // stream, then make the caller's stream wait on them.
      if (!copy_stream_) {
        copy_stream_ = std::make_unique<at::cuda::CUDAStream>(
            at::cuda::getStreamFromPool(false, output_device_->index()));
      }
      at::cuda::CUDAStream current =
          at::cuda::getCurrentCUDAStream(output_device_->index());
      {
        at::cuda::CUDAStreamGuard guard(*copy_stream_);
        batched = batch_nests_to_device(ptrs, batch_dim_, *output_device_);
      }
      auto ev = std::make_shared<at::cuda::CUDAEvent>();
      ev->record(*copy_stream_);
      ev->block(current);
      // The batch blocks were allocated on the copy stream but are consumed
      // (and eventually freed) on the caller's stream: tell the caching
      // allocator, or it may recycle them while the consumer still reads.
      batched.for_each([&current](const torch::Tensor& t) {
        t.record_stream(current);
      });
      if (source_pool_) {
        // Slab-pooled rollout slots recycle once the H2D copies complete.
        for (const auto* n : ptrs) {
          n->for_each([this, &ev](const torch::Tensor& t) {
            source_pool_->mark_consumed(t.data_ptr(), ev);
          });
        }
      }
    } else {
      batched = batch_nests(ptrs, batch_dim_);
      if (source_pool_) {
        for (const auto* n : ptrs) {
          n->for_each([this](const torch::Tensor& t) {
            source_pool_->mark_consumed(t.data_ptr(), nullptr);
          });
        }
      }
    }
    auto result = std::make_pair(std::move(batched),
                                 static_cast<int64_t>(items.size()));
    wait_stats_.add(t1 - t0);
    cat_stats_.add(now_us() - t1);
    return result;
  }

  int64_t size() const { return queue_.size(); }
  void close() { queue_.close(); }
  bool is_closed() const { return queue_.is_closed(); }

  std::map<std::string, double> stats() const {
    std::map<std::string, double> out;
    out["dequeues"] = wait_stats_.n.load();
    if (wait_stats_.n > 0) {
      out["avg_wait_ms"] = wait_stats_.sum_us / 1e3 / wait_stats_.n;
      out["avg_cat_ms"] = cat_stats_.sum_us / 1e3 / cat_stats_.n;
    }
    if (source_pool_) {
      for (const auto& kv : source_pool_->stats()) out[kv.first] = kv.second;
    }
    return out;
  }

  void set_source_pool(std::shared_ptr<PinnedSlabPool> pool) {
    source_pool_ = std::move(pool);
  }

  // Drop accumulated latency stats (steady-state measurement after warmup).
  void reset_stats() {
    wait_stats_.reset();
    cat_stats_.reset();
  }

 private:
  const std::optional<torch::Device> output_device_;
  const int64_t batch_dim_;
  const int64_t min_batch_size_;
  const int64_t max_batch_size_;
  const std::optional<std::chrono::milliseconds> timeout_;
  const bool check_inputs_;
  BoundedQueue<TensorNest> queue_;
  std::shared_ptr<PinnedSlabPool> source_pool_;
  std::unique_ptr<at::cuda::CUDAStream> copy_stream_;
  mutable StageStats wait_stats_;
  mutable StageStats cat_stats_;
};

// ---------------------------------------------------------------------------
// DynamicBatcher: many blocking compute() callers -> one batched inference.
// ---------------------------------------------------------------------------

class DynamicBatcher {
 public:
  struct Request {
    TensorNest inputs;
    int64_t batch_size;  // along batch_dim
    std::shared_ptr<std::promise<TensorNest>> promise;
  };

  class Batch {
   public:
    Batch(int64_t batch_dim, std::vector<Request> requests, bool check_outputs,
          StageStats* service_stats = nullptr)
        : batch_dim_(batch_dim),
          requests_(std::move(requests)),
          check_outputs_(check_outputs),
          service_stats_(service_stats),
          created_us_(now_us()) {}

    ~Batch() {
      if (!fulfilled_) {
        auto eptr = std::make_exception_ptr(
            AsyncError("Batch destroyed before set_outputs was called"));
        // fulfilled_count_ promises already carry values (set_outputs threw
        // mid-way); breaking those again would raise future_error out of a
        // destructor, so only fail the rest.
        for (size_t i = fulfilled_count_; i < requests_.size(); ++i) {
          requests_[i].promise->set_exception(eptr);
        }
      }
    }

    int64_t size() const {
      int64_t total = 0;
      for (const auto& r : requests_) total += r.batch_size;
      return total;
    }

    TensorNest get_inputs() {
      std::vector<const TensorNest*> ptrs;
      ptrs.reserve(requests_.size());
      for (const auto& r : requests_) ptrs.push_back(&r.inputs);
      return batch_nests(ptrs, batch_dim_);
    }

    void set_outputs(TensorNest outputs) {
      if (fulfilled_) {
        throw std::runtime_error("set_outputs called twice");
      }
      if (check_outputs_) {
        const int64_t expected = size();
        outputs.for_each([this, expected](const torch::Tensor& t) {
          if (t.dim() <= batch_dim_) {
            throw std::invalid_argument(
                "With batch_dim == " + std::to_string(batch_dim_) +
                ", output shape must have at least " +
                std::to_string(batch_dim_ + 1) + " dims but got " +
                std::to_string(t.dim()));
          }
          if (t.size(batch_dim_) != expected) {
            throw std::invalid_argument(
                "Output shape must have the same batch dimension as the "
                "input batch size. Expected: " + std::to_string(expected) +
                ". Observed: " + std::to_string(t.size(batch_dim_)));
          }
        });
      }
      // Slice for every caller BEFORE fulfilling any promise: narrow() can
      // throw on a mis-shaped output (when check_outputs is off), and a
      // half-fulfilled batch would make the destructor re-break satisfied
      // promises.
      std::vector<TensorNest> slices;
      slices.reserve(requests_.size());
      int64_t offset = 0;
      for (auto& req : requests_) {
        slices.push_back(
            outputs.map([this, offset, &req](const torch::Tensor& t) {
              return t.narrow(batch_dim_, offset, req.batch_size);
            }));
        offset += req.batch_size;
      }
      for (size_t i = 0; i < requests_.size(); ++i) {
        requests_[i].promise->set_value(std::move(slices[i]));
        fulfilled_count_ = i + 1;
      }
      fulfilled_ = true;
      if (service_stats_ != nullptr) {
        service_stats_->add(now_us() - created_us_);
      }
    }

   private:
    const int64_t batch_dim_;
    std::vector<Request> requests_;
    const bool check_outputs_;
    StageStats* service_stats_;
    const int64_t created_us_;
    bool fulfilled_ = false;
    size_t fulfilled_count_ = 0;
  };

  DynamicBatcher(int64_t batch_dim = 0,
                 std::optional<int64_t> minimum_batch_size = std::nullopt,
                 std::optional<int64_t> maximum_batch_size = std::nullopt,
                 std::optional<int64_t> timeout_ms = std::nullopt,
                 bool check_outputs = true)
      : batch_dim_(batch_dim),
        min_batch_size_(minimum_batch_size ? *minimum_batch_size : 1),
        max_batch_size_(maximum_batch_size
                            ? *maximum_batch_size
                            : std::numeric_limits<int64_t>::max()),
        timeout_(timeout_ms
                     ? std::optional<std::chrono::milliseconds>(
                           std::chrono::milliseconds(*timeout_ms))
                     : std::nullopt),
        check_outputs_(check_outputs),
        queue_(std::nullopt) {}

  // Called by actor threads; blocks until the consumer sets outputs.
  // Enqueue a request and return the future (the multi-env actor loop
  // waits on several of these at once).
  std::future<TensorNest> compute_async(TensorNest inputs) {
    if (inputs.empty()) {
      throw std::invalid_argument("compute() on empty nest");
    }
    int64_t batch_size = -1;
    inputs.for_each([this, &batch_size](const torch::Tensor& t) {
      if (t.dim() <= batch_dim_) {
        throw std::invalid_argument("Input needs more dims than batch_dim");
      }
      if (batch_size < 0) {
        batch_size = t.size(batch_dim_);
      } else if (t.size(batch_dim_) != batch_size) {
        throw std::invalid_argument(
            "Input tensors disagree on the batch dimension");
      }
    });
    auto promise = std::make_shared<std::promise<TensorNest>>();
    std::future<TensorNest> future = promise->get_future();
    queue_.enqueue(Request{std::move(inputs), batch_size, std::move(promise)});
    return future;
  }

  TensorNest compute(TensorNest inputs) {
    const int64_t t0 = now_us();
    std::future<TensorNest> future = compute_async(std::move(inputs));
    if (future.wait_for(std::chrono::minutes(10)) ==
        std::future_status::timeout) {
      throw AsyncError("compute() timed out after 10 minutes");
    }
    auto result = future.get();  // Rethrows AsyncError from a dropped batch.
    roundtrip_stats_.add(now_us() - t0);
    return result;
  }

  std::shared_ptr<Batch> get_batch() {
    const int64_t t0 = now_us();
    std::vector<Request> requests =
        queue_.dequeue_many(min_batch_size_, max_batch_size_, timeout_);
    formation_stats_.add(now_us() - t0);
    int64_t total = 0;
    for (const auto& r : requests) total += r.batch_size;
    batch_size_stats_.add(total);
    return std::make_shared<Batch>(batch_dim_, std::move(requests),
                                   check_outputs_, &service_stats_);
  }

  int64_t size() const { return queue_.size(); }
  void close() { queue_.close(); }
  bool is_closed() const { return queue_.is_closed(); }

  std::map<std::string, double> stats() const {
    std::map<std::string, double> out;
    out["computes"] = roundtrip_stats_.n.load();
    out["batches"] = batch_size_stats_.n.load();
    if (roundtrip_stats_.n > 0) {
      out["avg_roundtrip_ms"] =
          roundtrip_stats_.sum_us / 1e3 / roundtrip_stats_.n;
    }
    if (batch_size_stats_.n > 0) {
      out["avg_batch_size"] =
          double(batch_size_stats_.sum_us) / batch_size_stats_.n;
      out["avg_formation_wait_ms"] =
          formation_stats_.sum_us / 1e3 / formation_stats_.n;
    }
    if (service_stats_.n > 0) {
      out["avg_service_ms"] = service_stats_.sum_us / 1e3 / service_stats_.n;
    }
    return out;
  }

  void reset_stats() {
    roundtrip_stats_.reset();
    formation_stats_.reset();
    batch_size_stats_.reset();
    service_stats_.reset();
  }

 private:
  const int64_t batch_dim_;
  const int64_t min_batch_size_;
  const int64_t max_batch_size_;
  const std::optional<std::chrono::milliseconds> timeout_;
  const bool check_outputs_;
  BoundedQueue<Request> queue_;
  mutable StageStats roundtrip_stats_;
  mutable StageStats formation_stats_;
  mutable StageStats service_stats_;
  mutable StageStats batch_size_stats_;  // sum_us field holds summed sizes
};

}  // namespace tbruntime
