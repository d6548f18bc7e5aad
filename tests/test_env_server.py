"""End-to-end loopback: Python envs behind the unix-socket EnvServer, driven
by a real ActorPool + DynamicBatcher in-process (ref test strategy:
tests/core_agent_state_test.py, tests/contiguous_arrays_test.py)."""

import os
import threading
import uuid

import numpy as np
import pytest
import torch

from torchbeast_amd import runtime
from torchbeast_amd.envs.synthetic import CountingEnv


def _address():
    return f"unix:/tmp/tbamd-test-{uuid.uuid4().hex[:8]}.sock"


def _run_pool(unroll_length, addresses, inference_fn, n_rollouts,
              initial_agent_state=()):
    """Drive a real ActorPool against `addresses`; collect n_rollouts."""
    learner_queue = runtime.BatchingQueue(
        batch_dim=1, minimum_batch_size=1, maximum_batch_size=1
    )
    batcher = runtime.DynamicBatcher(batch_dim=1, minimum_batch_size=1,
                                     maximum_batch_size=64, timeout_ms=5)
    pool = runtime.ActorPool(
        unroll_length=unroll_length,
        learner_queue=learner_queue,
        inference_batcher=batcher,
        env_server_addresses=addresses,
        initial_agent_state=initial_agent_state,
    )
    pool_thread = threading.Thread(target=pool.run, daemon=True)
    pool_thread.start()

    def inference():
        try:
            for batch in batcher:
                batch.set_outputs(inference_fn(*batch.get_inputs()))
        except runtime.ClosedBatchingQueue:
            pass

    inf_thread = threading.Thread(target=inference, daemon=True)
    inf_thread.start()

    rollouts = []
    it = iter(learner_queue)
    for _ in range(n_rollouts):
        rollouts.append(next(it))
    batcher.close()
    learner_queue.close()
    pool_thread.join(5)
    inf_thread.join(5)
    return rollouts


def test_socket_env_roundtrip_counting():
    addr = _address()
    server = runtime.Server(lambda: CountingEnv(episode_length=5), addr)
    server.start()
    try:
        def policy(env_outputs, agent_state):
            frame = env_outputs[0]
            b = frame.shape[1]
            action = torch.zeros(1, b, dtype=torch.int64)
            logits = torch.zeros(1, b, 2)
            baseline = torch.zeros(1, b)
            return ((action, logits, baseline), agent_state)

        rollouts = _run_pool(4, [addr], policy, n_rollouts=3)
    finally:
        server.stop()

    # Frames count 0,1,2,... continuously; rollouts overlap by one step.
    frames = [r[0][0][0].flatten().tolist() for r in rollouts]
    assert frames[0] == [0, 1, 2, 3, 4]
    assert frames[1][0] == frames[0][-1]
    assert frames[1] == [4, 5, 6, 7, 8]

    # done fires when counter hits a multiple of 5 (plus the initial step),
    # and episode bookkeeping resets after it.
    env_outputs, agent_outputs = rollouts[0][0]
    frame, reward, done, episode_step, episode_return = env_outputs
    assert done.flatten().tolist() == [True, False, False, False, False]
    assert episode_step.flatten().tolist() == [0, 1, 2, 3, 4]
    assert reward.flatten().tolist() == [0.0, 1.0, 2.0, 3.0, 4.0]
    env_outputs1 = rollouts[1][0][0]
    assert env_outputs1[2].flatten().tolist() == [False, True, False, False, False]
    # episode_return at done reports the completed episode: 1+2+3+4+5.
    assert env_outputs1[4].flatten().tolist()[1] == 15.0


def test_recurrent_agent_state_across_batching():
    """The reference's core-agent-state invariant: the state stored with each
    rollout must equal the state before its first step's inference
    (ref: tests/core_agent_state_test.py:95-110)."""
    addr = _address()
    L = 5
    server = runtime.Server(lambda: CountingEnv(episode_length=L), addr)
    server.start()
    try:
        def policy(env_outputs, agent_state):
            frame, _, done, *_ = env_outputs
            b = frame.shape[1]
            (state,) = agent_state
            # Zero on done, then count this step.
            new_state = state * (~done).float().view(1, b, 1) + 1
            action = torch.zeros(1, b, dtype=torch.int64)
            logits = torch.zeros(1, b, 2)
            baseline = torch.zeros(1, b)
            return ((action, logits, baseline), (new_state,))

        rollouts = _run_pool(
            3, [addr], policy, n_rollouts=4,
            initial_agent_state=(torch.zeros(1, 1, 1),),
        )
    finally:
        server.stop()

    def oracle_state(f):
        # Steps since episode start, before masking happens inside compute.
        if f == 0:
            return 0.0
        return float(L) if f % L == 0 else float(f % L)

    for rollout in rollouts:
        (env_outputs, _agent_outputs), (init_state,) = rollout
        first_frame = int(env_outputs[0].flatten()[0])
        assert float(init_state.flatten()[0]) == oracle_state(first_frame)


class TransposedObsEnv:
    """Observations are transposes of a C-order array (non-contiguous)."""

    def __init__(self):
        self._t = 0

    def _obs(self):
        base = np.arange(12, dtype=np.float32).reshape(3, 4) + 100 * self._t
        return base.T  # (4, 3), non-contiguous

    def reset(self):
        self._t = 0
        return self._obs()

    def step(self, action):
        self._t += 1
        return self._obs(), 0.0, False, {}


def test_noncontiguous_observations_arrive_intact():
    """ref: tests/contiguous_arrays_test.py — transposed arrays must survive
    serialization."""
    addr = _address()
    server = runtime.Server(TransposedObsEnv, addr)
    server.start()
    try:
        def policy(env_outputs, agent_state):
            b = env_outputs[0].shape[1]
            return (
                (torch.zeros(1, b, dtype=torch.int64), torch.zeros(1, b, 2),
                 torch.zeros(1, b)),
                agent_state,
            )

        rollouts = _run_pool(2, [addr], policy, n_rollouts=1)
    finally:
        server.stop()

    frames = rollouts[0][0][0][0]
    assert frames.shape == (3, 1, 4, 3)
    expected0 = np.arange(12, dtype=np.float32).reshape(3, 4).T
    np.testing.assert_array_equal(frames[0, 0].numpy(), expected0)
    expected1 = expected0 + 100
    np.testing.assert_array_equal(frames[1, 0].numpy(), expected1)


def test_server_socket_file_cleanup():
    addr = _address()
    path = addr[len("unix:"):]
    server = runtime.Server(lambda: CountingEnv(), addr)
    server.start()
    assert os.path.exists(path)
    server.stop()
    assert not os.path.exists(path)


class MultiDtypeEnv:
    """Fixture covering the wire codec's dtype table: one observation array
    of the parametrized dtype, offset per step so values are checkable."""

    def __init__(self, dtype):
        self.dtype = dtype
        self._t = 0

    def reset(self):
        self._t = 0
        return np.arange(6, dtype=self.dtype).reshape(2, 3)

    def step(self, action):
        self._t += 1
        obs = (np.arange(6, dtype=self.dtype).reshape(2, 3)
               + np.asarray(self._t, dtype=self.dtype))
        return obs, 0.5, False, {}


@pytest.mark.parametrize("dtype", [np.uint8, np.int16, np.int32, np.int64,
                                   np.float32, np.float64])
def test_wire_roundtrip_dtypes(dtype):
    addr = _address()
    server = runtime.Server(lambda: MultiDtypeEnv(dtype), addr)
    server.start()
    try:
        def policy(env_outputs, agent_state):
            b = env_outputs[0].shape[1]
            return ((torch.zeros(1, b, dtype=torch.int64),
                     torch.zeros(1, b, 2), torch.zeros(1, b)), agent_state)

        rollouts = _run_pool(2, [addr], policy, n_rollouts=1)
    finally:
        server.stop()
    frames = rollouts[0][0][0][0]  # [T+1, 1, 2, 3]
    expected0 = np.arange(6, dtype=dtype).reshape(2, 3)
    np.testing.assert_array_equal(frames[0, 0].numpy(), expected0)
    np.testing.assert_array_equal(frames[1, 0].numpy(), expected0 + 1)


def test_tcp_env_roundtrip():
    """Cross-machine-capable env plane: the same framed protocol over TCP
    (the reference's gRPC plane worked over any channel; wire.h + tcp:
    addresses restore that capability)."""
    import socket as pysocket

    s = pysocket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()

    address = f"tcp:127.0.0.1:{port}"
    server = runtime.Server(lambda: CountingEnv(episode_length=5), address)
    server.start()
    try:
        def inference(env_outputs, agent_state):
            frame = env_outputs[0]
            b = frame.shape[1]
            action = torch.zeros((1, b), dtype=torch.int64)
            logits = torch.ones((1, b, 2))
            baseline = torch.zeros((1, b))
            return ((action, logits, baseline), agent_state)

        rollouts = _run_pool(unroll_length=6, addresses=[address],
                             inference_fn=inference, n_rollouts=2)
        (env_outputs, _), _ = rollouts[0]
        frames = env_outputs[0]
        # CountingEnv counts 0..limit; rollout frames must follow it.
        assert frames.shape[0] == 7  # T+1
    finally:
        server.stop()


def test_obs_slab_slot_requests():
    """Slot-id request mode: actors publish observations into the pinned
    slab; inference sees only slot ids. (On GPU the C++ engine gathers the
    slab device-side; here a Python closure reads it directly.)"""
    L = 6
    learner_queue = runtime.BatchingQueue(
        batch_dim=1, minimum_batch_size=1, maximum_batch_size=1
    )
    batcher = runtime.DynamicBatcher(batch_dim=1, minimum_batch_size=1,
                                     maximum_batch_size=64, timeout_ms=5)
    pool = runtime.ActorPool(
        unroll_length=4,
        learner_queue=learner_queue,
        inference_batcher=batcher,
        env_server_addresses=["synthetic:1x8x16:3:%d" % L] * 3,
        initial_agent_state=(),
        use_obs_slab=True,
    )
    pool_thread = threading.Thread(target=pool.run, daemon=True)
    pool_thread.start()
    slab_frames, slab_rew, slab_done = pool.obs_slab()
    assert slab_frames.shape == (3, 1, 8, 16)
    if torch.cuda.is_available():
        assert slab_frames.is_pinned()
    seen_ids = set()

    def inference():
        try:
            for batch in batcher:
                ids, agent_state = batch.get_inputs()
                assert ids.dtype == torch.int32 and ids.dim() == 2
                b = ids.shape[1]
                for i in ids[0].tolist():
                    seen_ids.add(i)
                action = torch.zeros((1, b), dtype=torch.int64)
                logits = torch.ones((1, b, 3))
                baseline = torch.zeros((1, b))
                batch.set_outputs(((action, logits, baseline), agent_state))
        except runtime.ClosedBatchingQueue:
            pass

    inf_thread = threading.Thread(target=inference, daemon=True)
    inf_thread.start()
    it = iter(learner_queue)
    rollouts = [next(it) for _ in range(3)]
    batcher.close()
    learner_queue.close()
    pool_thread.join(5)
    inf_thread.join(5)
    # All three actors used their own slot; rollouts still carry full frames.
    assert seen_ids == {0, 1, 2}
    (env_outputs, _agent), _state = rollouts[0]
    assert env_outputs[0].shape[0] == 5  # [T+1, 1, ...] frames


def test_multi_env_actor_threads():
    """Event-driven actor mode: one thread drives several env streams with
    overlapped inference futures; rollout contents must match the
    single-env-per-thread contract (CountingEnv frame continuity)."""
    L = 10
    learner_queue = runtime.BatchingQueue(
        batch_dim=1, minimum_batch_size=1, maximum_batch_size=1
    )
    batcher = runtime.DynamicBatcher(batch_dim=1, minimum_batch_size=1,
                                     maximum_batch_size=64, timeout_ms=3)
    pool = runtime.ActorPool(
        unroll_length=4,
        learner_queue=learner_queue,
        inference_batcher=batcher,
        env_server_addresses=["synthetic:1x8x16:3:%d" % L] * 6,
        initial_agent_state=(),
        envs_per_thread=3,  # 6 envs on 2 threads
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
    rollouts = [next(it) for _ in range(8)]
    batcher.close()
    learner_queue.close()
    pool_thread.join(5)
    for (env_outputs, _agent), _state in rollouts:
        frames = env_outputs[0]  # [T+1, 1, 1, 8, 16]
        assert frames.shape[0] == 5
        # Synthetic env frames vary per step; the rollout must be a
        # contiguous slice of ONE env stream: overlapping rollouts from
        # different envs would show duplicated constant frames.
        vals = frames[:, 0, 0, 0, 0]
        assert len(set(vals.tolist())) >= 2
