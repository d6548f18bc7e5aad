"""Data-parallel pieces over gloo, world_size 2, single machine (no GPU
needed; the same code paths run over RCCL on MI355X)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from torchbeast_amd.parallel import ddp as tbddp
from torchbeast_amd.parallel import flat as tbflat


def _find_free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker_allreduce(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)

        flat_grad = torch.full((10,), float(rank + 1))
        reducer = tbddp.GradAllReducer(flat_grad, world)
        reducer.reduce()
        # mean of (1, 2) = 1.5 everywhere.
        q.put((rank, flat_grad.tolist()))

        flat = torch.full((4,), float(rank * 7))
        tbddp.broadcast_flat(flat, src=0)
        q.put((rank + 100, flat.tolist()))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put(("error", repr(e)))


@pytest.mark.timeout(120)
def test_grad_allreduce_and_broadcast_world2():
    port = _find_free_port()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [
        ctx.Process(target=_worker_allreduce, args=(r, 2, port, q))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(4):
        k, v = q.get()
        assert k != "error", v
        results[k] = v
    for p in procs:
        p.join(30)

    assert results[0] == results[1] == [1.5] * 10
    assert results[100] == results[101] == [0.0] * 4


def _worker_train_step(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.manual_seed(7)  # Same init on both ranks.
        net = torch.nn.Linear(4, 2)
        flat_param = tbflat.flatten_parameters(net)
        flat_grad = tbflat.attach_flat_grads(net)
        opt = tbflat.FusedRMSProp(flat_param, flat_grad, lr=0.05, clip_norm=10.0)
        reducer = tbddp.GradAllReducer(flat_grad, world)

        # Different data per rank; identical params expected after each step.
        for step in range(3):
            torch.manual_seed(1000 * rank + step)
            x = torch.randn(6, 4)
            opt.zero_grad()
            net(x).pow(2).sum().backward()
            reducer.reduce()
            opt.step()
        q.put((rank, flat_param.tolist()))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put(("error", repr(e)))


@pytest.mark.timeout(120)
def test_dp_training_keeps_replicas_identical():
    port = _find_free_port()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [
        ctx.Process(target=_worker_train_step, args=(r, 2, port, q))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        k, v = q.get()
        assert k != "error", v
        results[k] = v
    for p in procs:
        p.join(30)
    assert results[0] == pytest.approx(results[1])
