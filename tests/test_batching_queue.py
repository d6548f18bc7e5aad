"""BatchingQueue semantics and stress (ref test strategy:
tests/batching_queue_test.py)."""

import threading

import pytest
import torch

from torchbeast_amd import runtime


def test_constructor_validation():
    with pytest.raises(ValueError):
        runtime.BatchingQueue(minimum_batch_size=0)
    with pytest.raises(ValueError):
        runtime.BatchingQueue(minimum_batch_size=4, maximum_batch_size=2)
    with pytest.raises(ValueError):
        runtime.BatchingQueue(maximum_queue_size=0)


def test_close_semantics():
    q = runtime.BatchingQueue()
    q.close()
    with pytest.raises(runtime.ClosedBatchingQueue):
        q.close()
    with pytest.raises(runtime.ClosedBatchingQueue):
        q.enqueue(torch.ones(1))
    assert q.is_closed()


def test_iteration_stops_on_close():
    q = runtime.BatchingQueue(batch_dim=0, minimum_batch_size=1, maximum_batch_size=1)
    q.enqueue(torch.ones(1, 2))
    q.close()
    items = list(q)
    assert len(items) == 1


def test_enqueue_requires_enough_dims():
    q = runtime.BatchingQueue(batch_dim=1)
    with pytest.raises(Exception):
        q.enqueue(torch.ones(3))  # 1-D tensor can't batch along dim 1.


def test_batches_concatenate_in_order():
    q = runtime.BatchingQueue(batch_dim=0, minimum_batch_size=4, maximum_batch_size=4)
    for i in range(4):
        q.enqueue({"x": torch.full((1, 2), float(i))})
    batch = next(iter(q))
    assert batch["x"].shape == (4, 2)
    torch.testing.assert_close(batch["x"][:, 0], torch.arange(4.0))


def test_maximum_queue_size_blocks_enqueue():
    q = runtime.BatchingQueue(
        batch_dim=0, minimum_batch_size=1, maximum_batch_size=1, maximum_queue_size=1
    )
    q.enqueue(torch.ones(1, 1))
    blocked = threading.Event()
    passed = threading.Event()

    def producer():
        blocked.set()
        q.enqueue(torch.ones(1, 1))  # Blocks until a dequeue frees a slot.
        passed.set()

    t = threading.Thread(target=producer)
    t.start()
    blocked.wait(2)
    assert not passed.wait(0.2)
    next(iter(q))
    assert passed.wait(2)
    t.join()


def test_producer_consumer_stress():
    n_producers, items_each = 8, 50
    q = runtime.BatchingQueue(batch_dim=0, minimum_batch_size=1, maximum_batch_size=16)
    total = n_producers * items_each
    seen = []
    seen_lock = threading.Lock()

    def producer(pid):
        for i in range(items_each):
            q.enqueue(torch.full((1, 1), float(pid * items_each + i)))

    def consumer():
        try:
            for batch in q:
                with seen_lock:
                    seen.extend(batch.flatten().tolist())
        except runtime.ClosedBatchingQueue:
            pass

    consumers = [threading.Thread(target=consumer) for _ in range(4)]
    for c in consumers:
        c.start()
    producers = [threading.Thread(target=producer, args=(p,)) for p in range(n_producers)]
    for p in producers:
        p.start()
    for p in producers:
        p.join()
    # Wait for drain, then close.
    import time

    deadline = time.time() + 10
    while time.time() < deadline:
        with seen_lock:
            if len(seen) == total:
                break
        time.sleep(0.01)
    q.close()
    for c in consumers:
        c.join()

    assert len(seen) == total
    assert sorted(int(v) for v in seen) == list(range(total))


def test_rollout_slab_ring_backpressure_and_recycling():
    """Budget-bounded pinned rollout ring: a tiny budget forces the pool to
    its minimum slot count, slots recycle through the learner queue's
    dequeue, and every rollout still arrives intact (conservation)."""
    import threading

    import torch

    from torchbeast_amd import runtime

    learner_queue = runtime.BatchingQueue(
        batch_dim=1, minimum_batch_size=2, maximum_batch_size=2,
        maximum_queue_size=4,
    )
    batcher = runtime.DynamicBatcher(batch_dim=1, minimum_batch_size=1,
                                     maximum_batch_size=64, timeout_ms=2)
    pool = runtime.ActorPool(
        unroll_length=3,
        learner_queue=learner_queue,
        inference_batcher=batcher,
        env_server_addresses=["synthetic:1x8x16:3:50"] * 4,
        initial_agent_state=(),
        rollout_budget_mb=1,  # floors at the 64-slot minimum
    )
    pool_thread = threading.Thread(target=pool.run, daemon=True)
    pool_thread.start()

    def inference():
        try:
            for batch in batcher:
                (frame, *_), agent_state = batch.get_inputs()
                b = frame.shape[1]
                batch.set_outputs(((torch.zeros((1, b), dtype=torch.int64),
                                    torch.ones((1, b, 3)),
                                    torch.zeros((1, b))), agent_state))
        except runtime.ClosedBatchingQueue:
            pass

    threading.Thread(target=inference, daemon=True).start()

    it = iter(learner_queue)
    got = 0
    for _ in range(20):
        (env_outputs, agent_outputs), _state = next(it)
        assert env_outputs[0].shape[:2] == (4, 2)  # [T+1, batch=2, ...]
        got += 2
    stats = learner_queue.stats()
    assert stats["slab_slots"] >= 64
    assert stats["slab_slot_bytes"] > 0
    assert stats["slab_free"] <= stats["slab_slots"]
    batcher.close()
    learner_queue.close()
    pool_thread.join(5)
    assert got == 40
