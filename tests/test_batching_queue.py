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
