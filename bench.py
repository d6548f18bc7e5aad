"""Flagship benchmark: polybeast IMPALA training SPS on MI355X.

Measures the BASELINE.json metric — env-steps/sec (SPS), whole node,
AtariNet 84x84x4, unroll=80 — on synthetic Atari-shaped frames with
random-init weights, running the full actor-learner pipeline per GPU:
native synthetic envs -> C++ ActorPool -> DynamicBatcher -> batched GPU
inference -> rollout BatchingQueue -> learner step (V-trace + fused IMPALA
loss + fused clip/RMSProp), with a flat-gradient RCCL all-reduce across
ranks when N > 1.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  # N>1: python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
  #        --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

A "step" is one learner update consuming unroll_length*batch_size env steps.
Rank 0 prints ONE JSON line with the whole-job aggregate SPS.
"""

import argparse
import json
import os
import sys
import threading
import timeit

import torch

import torchbeast_amd.polybeast_learner as pbl
from torchbeast_amd import runtime
from torchbeast_amd.core import vtrace
from torchbeast_amd.ops import functional as tbops
from torchbeast_amd.parallel import ddp as tbddp
from torchbeast_amd.parallel import flat as tbflat

METRIC = (
    "env-steps/sec (SPS) whole node, AtariNet 84×84×4 unroll=80 "
    "at 1/2/4/8 MI355X"
)


def parse_args():
    p = argparse.ArgumentParser(description="torchbeast_amd flagship benchmark")
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch_size", type=int, default=32)
    p.add_argument("--unroll_length", type=int, default=80)
    p.add_argument("--actors", type=int, default=640,
                   help="Env streams per GPU (tuned on MI355X; see "
                        "profiles/PROFILE_r2.md).")
    p.add_argument("--model", default="shallow", choices=["shallow", "deep"])
    p.add_argument("--use_lstm", action="store_true")
    p.add_argument("--frame", default="4x84x84",
                   help="Synthetic frame shape CxHxW.")
    p.add_argument("--num_actions", type=int, default=6)
    p.add_argument("--num_inference_threads", type=int, default=8)
    p.add_argument("--inference_min_batch_size", type=int, default=128)
    p.add_argument("--inference_max_batch_size", type=int, default=512)
    p.add_argument("--inference_timeout_ms", type=int, default=5)
    p.add_argument("--episode_length", type=int, default=1000)
    p.add_argument("--rollout_budget_mb", type=int, default=1024)
    p.add_argument("--envs_per_thread", type=int, default=16,
                   help="Env streams per actor thread (event-driven loop; "
                        "512 one-env threads measured host-scheduling "
                        "bound).")
    p.add_argument("--hipgraph", action="store_true", default=True,
                   help="Capture the learner step in a hipGraph (1-GPU).")
    p.add_argument("--no_hipgraph", dest="hipgraph", action="store_false")
    p.add_argument("--py_inference", action="store_true",
                   help="Python inference threads instead of the C++ engine.")
    return p.parse_args()


def learner_step(flags, batch_tensors, model, optimizer, scheduler, reducer,
                 actor_flat, flat_param, inference_runner=None,
                 device_ops_only=False):
    env_outputs = pbl.EnvOutput._make(batch_tensors[:5])
    actor_outputs = pbl.AgentOutput._make(batch_tensors[5:8])
    initial_agent_state = batch_tensors[8:]

    learner_outputs, _ = model(
        dict(frame=env_outputs.frame, reward=env_outputs.rewards,
             done=env_outputs.done),
        initial_agent_state,
    )
    learner_outputs = pbl.AgentOutput._make(pbl._as_agent_output(learner_outputs))
    bootstrap_value = learner_outputs.baseline[-1]

    env_outputs = pbl.EnvOutput._make(t[1:] for t in env_outputs)
    actor_outputs = pbl.AgentOutput._make(t[1:] for t in actor_outputs)
    learner_outputs = pbl.AgentOutput._make(t[:-1] for t in learner_outputs)

    clipped_rewards = torch.clamp(env_outputs.rewards, -1, 1)
    discounts = (~env_outputs.done).float() * flags.discounting

    vtr = vtrace.from_logits(
        behavior_policy_logits=actor_outputs.policy_logits,
        target_policy_logits=learner_outputs.policy_logits,
        actions=actor_outputs.action,
        discounts=discounts,
        rewards=clipped_rewards,
        values=learner_outputs.baseline,
        bootstrap_value=bootstrap_value,
    )
    pg_loss, baseline_loss, entropy_loss = tbops.fused_impala_loss(
        learner_outputs.policy_logits,
        learner_outputs.baseline,
        actor_outputs.action,
        vtr.pg_advantages,
        vtr.vs,
    )
    total_loss = (pg_loss + flags.baseline_cost * baseline_loss
                  + flags.entropy_cost * entropy_loss)

    optimizer.zero_grad()
    total_loss.backward()
    reducer.reduce()
    optimizer.step()
    with torch.no_grad():
        actor_flat.copy_(flat_param)
    if device_ops_only:
        # hipGraph capture: host-side updates (LR schedule, runner weight
        # notification) happen outside the captured region, per replay.
        return total_loss
    scheduler.step()
    if inference_runner is not None:
        inference_runner.mark_weights_dirty()
    return total_loss


def main():
    args = parse_args()
    rank, world_size, local_rank = tbddp.maybe_init_distributed()

    # Distinct RNG per rank (sampling, init noise); weights are broadcast
    # from rank 0 afterwards, so replicas still start identical.
    torch.manual_seed(4242 + rank * 977)

    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda", local_rank) if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    # Reuse the polybeast flag namespace for model/loss hyperparameters.
    flags = pbl.parser.parse_args([])
    flags.env = f"synthetic:{args.frame}:{args.num_actions}:{args.episode_length}"
    flags.model = args.model
    flags.use_lstm = args.use_lstm
    flags.num_actions = args.num_actions
    flags.batch_size = args.batch_size
    flags.unroll_length = args.unroll_length
    flags.learner_device = device
    flags.actor_device = device

    T, B = args.unroll_length, args.batch_size

    model = pbl.create_model(flags).to(device)
    actor_model = pbl.create_model(flags).to(device)
    flat_param = tbflat.flatten_parameters(model)
    flat_grad = tbflat.attach_flat_grads(model)
    actor_flat = tbflat.flatten_parameters(actor_model)
    tbddp.broadcast_flat(flat_param)
    with torch.no_grad():
        actor_flat.copy_(flat_param)

    optimizer = tbflat.FusedRMSProp(
        flat_param, flat_grad, lr=flags.learning_rate, alpha=flags.alpha,
        eps=flags.epsilon, clip_norm=flags.grad_norm_clipping,
    )
    total_env_steps = (args.warmup + args.steps) * T * B * world_size * 10
    scheduler = tbflat.LinearLR(optimizer, T * B * world_size, total_env_steps)
    reducer = tbddp.GradAllReducer(
        flat_grad, world_size,
        params=[p for p in model.parameters() if p.requires_grad])

    learner_queue = runtime.BatchingQueue(
        batch_dim=1, minimum_batch_size=B, maximum_batch_size=B,
        maximum_queue_size=B,
        output_device=str(device) if use_cuda else None,
    )
    inference_batcher = runtime.DynamicBatcher(
        batch_dim=1, minimum_batch_size=args.inference_min_batch_size,
        maximum_batch_size=args.inference_max_batch_size,
        timeout_ms=args.inference_timeout_ms,
    )
    addresses = [flags.env] * args.actors
    initial_agent_state = tuple(
        t.cpu() for t in model.initial_state(batch_size=1)
    )
    use_cpp_inference = use_cuda and not args.py_inference
    # Measured on MI355X: for 84x84 frames the pinned-cat + SDMA path wins
    # (~12%) over the GPU-side slab gather, but for full-res frames the
    # per-batch pinned cat explodes (26+ ms/batch at 3x210x160) and the
    # slot-id slab path wins by an order of magnitude. Threshold on frame
    # bytes; TBAMD_OBS_SLAB=1 / TBAMD_NO_OBS_SLAB=1 force either way.
    frame_bytes = 1
    for d in (int(x) for x in args.frame.split("x")):
        frame_bytes *= d
    use_obs_slab = use_cpp_inference and not os.environ.get(
        "TBAMD_NO_OBS_SLAB") and (
        bool(os.environ.get("TBAMD_OBS_SLAB")) or frame_bytes > 48 * 1024)
    pool = runtime.ActorPool(
        unroll_length=T, learner_queue=learner_queue,
        inference_batcher=inference_batcher,
        env_server_addresses=addresses,
        initial_agent_state=initial_agent_state,
        seed_base=rank * args.actors,  # distinct env streams per rank
        use_obs_slab=use_obs_slab,
        rollout_budget_mb=args.rollout_budget_mb,
        envs_per_thread=args.envs_per_thread,
    )
    def run_pool():
        try:
            pool.run()
        except Exception:
            import traceback
            traceback.print_exc()
            # Make actor death loud: unblock the learner loop so the bench
            # fails fast instead of hanging on an empty queue.
            try:
                inference_batcher.close()
            except Exception:
                pass
            try:
                learner_queue.close()
            except Exception:
                pass

    pool_thread = threading.Thread(target=run_pool, daemon=True)
    pool_thread.start()

    inference_runner = None
    if use_cpp_inference:
        inference_runner = pbl.make_inference_runner(
            actor_model, inference_batcher
        )
        if use_obs_slab:
            slab = pool.obs_slab()  # blocks until the first env observation
            if slab:
                inference_runner.set_obs_slab(*slab)
        inference_runner.start(args.num_inference_threads)
    else:
        for _ in range(args.num_inference_threads):
            threading.Thread(
                target=pbl.inference,
                args=(flags, inference_batcher, actor_model),
                daemon=True,
            ).start()

    def get_batch(it):
        tensors = runtime._tbruntime.flatten(next(it))
        return tuple(t.to(device, non_blocking=True) for t in tensors)

    queue_iter = iter(learner_queue)

    graph = None
    static_batch = None
    # Default hipGraph for the non-LSTM configs (the LSTM's cooperative
    # persistent kernels capture but replay ~3x slower). The deep model's
    # ~150-op trunk step gains the most: replay removes the per-op GIL
    # handoffs that actor threads otherwise inflate to >100 ms/step.
    use_graph = (args.hipgraph and use_cuda and world_size == 1
                 and (not args.use_lstm
                      or os.environ.get("TBAMD_GRAPH_LSTM") == "1"))

    # Warmup (fills the pipeline, compiles/caches kernels).
    for _ in range(args.warmup):
        learner_step(flags, get_batch(queue_iter), model, optimizer, scheduler,
                     reducer, actor_flat, flat_param, inference_runner)

    if use_graph:
      try:
        # Capture one whole learner step (fwd + V-trace + loss + backward +
        # fused RMSProp + behavior sync) into a hipGraph; per-iteration the
        # batch is copied into the static input slots and the graph
        # replayed. LR decay flows through a device-resident lr scalar.
        optimizer.enable_device_lr()
        static_batch = tuple(t.clone() for t in get_batch(queue_iter))
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                learner_step(flags, static_batch, model, optimizer, scheduler,
                             reducer, actor_flat, flat_param, None,
                             device_ops_only=True)
        torch.cuda.current_stream().wait_stream(side)
        graph = torch.cuda.CUDAGraph()
        # thread_local capture: the inference engine and queue threads keep
        # synchronizing their own streams while we capture ours.
        with torch.cuda.graph(graph, capture_error_mode="thread_local"):
            learner_step(flags, static_batch, model, optimizer, scheduler,
                         reducer, actor_flat, flat_param, None,
                         device_ops_only=True)
      except Exception as e:  # pragma: no cover - device-dependent
        print(f"hipGraph capture failed ({e!r}); falling back to eager",
              file=sys.stderr)
        graph = None
        torch.cuda.synchronize()

    def run_step():
        if graph is None:
            learner_step(flags, get_batch(queue_iter), model, optimizer,
                         scheduler, reducer, actor_flat, flat_param,
                         inference_runner)
            return
        batch = get_batch(queue_iter)
        for dst, src in zip(static_batch, batch):
            dst.copy_(src, non_blocking=True)
        optimizer.push_lr()
        graph.replay()
        scheduler.step()
        if inference_runner is not None:
            inference_runner.mark_weights_dirty()

    timings = None
    if os.environ.get("TBAMD_BENCH_TIMINGS"):
        from torchbeast_amd.core.prof import Timings

        timings = Timings()
        # Steady-state stats: drop the startup transients accumulated
        # during warmup (first-serve compiles, allocator growth).
        learner_queue.reset_stats()
        inference_batcher.reset_stats()

    if world_size > 1:
        torch.distributed.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    start = timeit.default_timer()

    for _ in range(args.steps):
        if timings is None:
            run_step()
        elif graph is not None:
            timings.reset()
            run_step()
            torch.cuda.synchronize()
            timings.time("graph_step")
        else:
            timings.reset()
            batch = get_batch(queue_iter)
            torch.cuda.synchronize()
            timings.time("get_batch")
            learner_step(flags, batch, model, optimizer, scheduler, reducer,
                         actor_flat, flat_param, inference_runner)
            torch.cuda.synchronize()
            timings.time("learn")

    if use_cuda:
        torch.cuda.synchronize()
    if world_size > 1:
        torch.distributed.barrier()
    elapsed = timeit.default_timer() - start

    # MAX elapsed over ranks (barriers make local elapsed ≈ max already, but
    # reduce explicitly for correctness).
    if world_size > 1:
        e = torch.tensor([elapsed], device=device if use_cuda else None)
        torch.distributed.all_reduce(e, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(e.item())

    if timings is not None and rank == 0:
        print("bench timings:" + timings.summary(), file=sys.stderr)
        print(f"inference stats: {pbl.INFERENCE_STATS}", file=sys.stderr)
        print(f"learner_queue stats: {learner_queue.stats()}", file=sys.stderr)
        print(f"batcher stats: {inference_batcher.stats()}", file=sys.stderr)

    inference_batcher.close()
    learner_queue.close()
    if inference_runner is not None and os.environ.get("TBAMD_CLEAN_EXIT"):
        inference_runner.stop()

    if rank == 0:
        total_env_steps = args.steps * T * B * world_size
        sps = total_env_steps / elapsed
        result = {
            "metric": METRIC,
            "value": sps,
            "unit": "env_steps/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "AtariNet" if args.model == "shallow" else "IMPALA-ResNet",
                "global_batch": B * world_size,
                "seq_len": T,
                "parallelism": f"dp{world_size}",
                "actors_per_gpu": args.actors,
                "use_lstm": args.use_lstm,
                "frame": args.frame,
            },
        }
        print(json.dumps(result), flush=True)

    if world_size > 1:
        torch.distributed.destroy_process_group()
    if os.environ.get("TBAMD_CLEAN_EXIT"):
        # Orderly exit (lets rocprofv3 flush its results); threads are
        # daemonic and unwind via the closed queues.
        pool_thread.join(timeout=5)
        sys.exit(0)
    os._exit(0)  # Actor/inference threads are daemonic; skip teardown races.


if __name__ == "__main__":
    main()
