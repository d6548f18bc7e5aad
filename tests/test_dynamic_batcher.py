"""DynamicBatcher semantics and stress (ref test strategy:
tests/dynamic_batcher_test.py)."""

import threading
import time

import pytest
import torch

from torchbeast_amd import runtime


def _compute_async(batcher, inputs, out, idx):
    def run():
        try:
            out[idx] = batcher.compute(inputs)
        except Exception as e:  # noqa: BLE001 - tests inspect the error
            out[idx] = e

    t = threading.Thread(target=run)
    t.start()
    return t


def test_batch_slices_outputs_per_caller():
    b = runtime.DynamicBatcher(batch_dim=1, minimum_batch_size=2, maximum_batch_size=4)
    out = {}
    threads = [
        _compute_async(b, (torch.full((1, 1, 3), float(i)),), out, i)
        for i in range(2)
    ]
    batch = next(iter(b))
    (inputs,) = batch.get_inputs()
    assert inputs.shape == (1, 2, 3)
    batch.set_outputs((inputs * 2,))
    for t in threads:
        t.join()
    for i in range(2):
        (result,) = out[i]
        torch.testing.assert_close(result, torch.full((1, 1, 3), 2.0 * i))


def test_timeout_returns_partial_batch():
    b = runtime.DynamicBatcher(
        batch_dim=1, minimum_batch_size=8, maximum_batch_size=8, timeout_ms=100
    )
    out = {}
    t = _compute_async(b, (torch.ones(1, 1),), out, 0)
    start = time.time()
    batch = next(iter(b))
    elapsed = time.time() - start
    assert batch.size() == 1
    assert 0.05 < elapsed < 5.0
    batch.set_outputs((torch.zeros(1, 1),))
    t.join()


def test_dropped_batch_raises_async_error():
    b = runtime.DynamicBatcher(batch_dim=1, minimum_batch_size=1)
    out = {}
    t = _compute_async(b, (torch.ones(1, 1),), out, 0)
    batch = next(iter(b))
    del batch  # Dropped without set_outputs -> broken promise.
    t.join()
    assert isinstance(out[0], runtime.AsyncError)


def test_output_shape_validation():
    b = runtime.DynamicBatcher(batch_dim=1, minimum_batch_size=1)
    out = {}
    t = _compute_async(b, (torch.ones(1, 1),), out, 0)
    batch = next(iter(b))
    with pytest.raises(Exception, match="batch dimension"):
        batch.set_outputs((torch.ones(1, 5),))
    batch.set_outputs((torch.ones(1, 1),))
    t.join()
    assert not isinstance(out[0], Exception)


def test_double_set_outputs_raises():
    b = runtime.DynamicBatcher(batch_dim=1, minimum_batch_size=1)
    out = {}
    t = _compute_async(b, (torch.ones(1, 1),), out, 0)
    batch = next(iter(b))
    batch.set_outputs((torch.ones(1, 1),))
    with pytest.raises(Exception, match="twice"):
        batch.set_outputs((torch.ones(1, 1),))
    t.join()


def test_compute_on_mismatched_batch_dims_raises():
    b = runtime.DynamicBatcher(batch_dim=1, minimum_batch_size=1)
    with pytest.raises(Exception):
        b.compute((torch.ones(1, 1), torch.ones(1, 2)))


def test_many_producers_one_consumer_stress():
    b = runtime.DynamicBatcher(batch_dim=0, minimum_batch_size=1, maximum_batch_size=64)
    n, per = 16, 25
    out = {}
    threads = []
    for i in range(n * per):
        threads.append(_compute_async(b, (torch.full((1, 1), float(i)),), out, i))

    served = 0
    for batch in b:
        inputs = batch.get_inputs()
        batch.set_outputs((inputs[0] + 1000,))
        served += batch.size()
        if served == n * per:
            b.close()
    for t in threads:
        t.join()
    assert len(out) == n * per
    for i in range(n * per):
        (result,) = out[i]
        assert float(result) == i + 1000
