"""PolyBeast learner (ref: torchbeast/polybeast_learner.py) — MI355X-native.

One process per GPU (launched directly for 1 GPU, or via
`python -m torch.distributed.run --nproc-per-node N` for data-parallel
training over RCCL/xGMI). Each rank owns:

- a C++ ActorPool driving `--num_actors` environment streams (unix-socket
  env servers, or in-process native synthetic envs for `--env synthetic*`),
- a DynamicBatcher feeding batched behavior-model inference on the GPU,
- a BatchingQueue of [T+1, B] rollouts feeding the learner step
  (V-trace + IMPALA losses + fused clip/RMSProp on a flat parameter buffer),
- with world_size > 1, a single flat-gradient RCCL all-reduce per step.

Differences from the reference runtime (same capabilities, MI355X design):
queues hand over pinned host tensors (DMA H2D), the optimizer/clip/LR are
one fused kernel over a flat buffer, and weight sync to the behavior model
is one flat device copy instead of a state_dict walk.
"""

import argparse
import collections
import logging
import os
import threading
import time
import timeit

import torch

from torchbeast_amd import flags as tbflags
from torchbeast_amd import runtime
from torchbeast_amd.core import file_writer, vtrace
from torchbeast_amd.models.atari_net import AtariNet
from torchbeast_amd.models.resnet import ResNet
from torchbeast_amd.ops import functional as tbops
from torchbeast_amd.parallel import ddp as tbddp
from torchbeast_amd.parallel import flat as tbflat

logging.basicConfig(
    format="[%(levelname)s:%(process)d %(module)s:%(lineno)d %(asctime)s] %(message)s",
    level=0,
)

EnvOutput = collections.namedtuple(
    "EnvOutput", "frame rewards done episode_step episode_return"
)
AgentOutput = collections.namedtuple("AgentOutput", "action policy_logits baseline")


def make_parser():
    parser = argparse.ArgumentParser(description="MI355X-native PolyBeast learner")
    parser.add_argument("--mode", default="train", choices=["train", "test"])
    tbflags.add_common_flags(parser)
    parser.add_argument("--pipes_basename", default="unix:/tmp/polybeast",
                        help="Basename for the env-server unix sockets.")
    parser.add_argument("--num_actors", default=4, type=int,
                        help="Env streams driven by this rank's ActorPool.")
    parser.add_argument("--num_learner_threads", default=2, type=int)
    parser.add_argument("--num_inference_threads", default=2, type=int)
    parser.add_argument("--max_learner_queue_size", default=None, type=int)
    parser.add_argument("--envs_per_thread", default=0, type=int,
                        help="Env streams per actor thread (0 = auto: 8 for "
                             "in-process synthetic envs, 1 for socket envs "
                             "so a slow remote env cannot stall neighbors).")
    parser.add_argument("--rollout_buffer_budget_mb", default=1024, type=int,
                        help="Pinned rollout ring budget (MB); actors block "
                             "when it is exhausted (0 = unbounded ad-hoc "
                             "pinned allocations).")
    parser.add_argument("--num_actions", default=6, type=int)
    parser.add_argument("--model", default="shallow", choices=["shallow", "deep"],
                        help="shallow=AtariNet (the headline bench model), "
                             "deep=IMPALA ResNet.")
    parser.add_argument("--inference_min_batch_size", default=1, type=int)
    parser.add_argument("--inference_max_batch_size", default=512, type=int)
    parser.add_argument("--inference_timeout_ms", default=10, type=int)
    parser.add_argument("--write_profiler_trace", action="store_true")
    parser.add_argument("--use_hipgraph", action="store_true",
                        help="Capture the learner step in a hipGraph "
                             "(1 GPU, 1 learner thread, shallow non-LSTM).")
    parser.add_argument("--py_inference", action="store_true",
                        help="Serve inference from Python threads instead of "
                             "the C++ engine (always the case for --model "
                             "deep or on CPU).")
    tbflags.add_loss_flags(parser)
    tbflags.add_optimizer_flags(parser)
    return parser


parser = make_parser()


def observation_shape(flags):
    spec = tbflags.parse_synthetic_env_spec(flags.env)
    if spec is not None:
        return spec[0], spec[1]
    return (4, 84, 84), flags.num_actions


def create_model(flags):
    shape, num_actions = observation_shape(flags)
    if flags.model == "shallow":
        return AtariNet(shape, num_actions, use_lstm=flags.use_lstm,
                        use_last_action=False)
    return ResNet(shape, num_actions, use_lstm=flags.use_lstm)


def _as_agent_output(outputs):
    """Normalize model output (AtariNet dict / ResNet tuple) to a tuple."""
    if isinstance(outputs, dict):
        return (outputs["action"], outputs["policy_logits"], outputs["baseline"])
    return tuple(outputs)


INFERENCE_STATS = {"batches": 0, "env_steps": 0}


def make_inference_runner(actor_model, inference_batcher, greedy=False):
    """GIL-free C++ inference engine over the behavior model's parameters
    (views of the flat actor buffer, so the learner's one-copy weight sync
    covers the runner too). Supports the shallow AtariNet and the deep
    IMPALA ResNet."""
    m = actor_model
    if hasattr(m, "feat_extract"):  # deep ResNet
        model_type = "deep"
        weights = []
        for section in m.feat_extract:
            weights += [section.conv.weight, section.conv.bias]
            for res in (section.res0, section.res1):
                weights += [res.conv0.weight, res.conv0.bias,
                            res.conv1.weight, res.conv1.bias]
    else:
        model_type = "shallow"
        weights = [
            m.conv1.weight, m.conv1.bias,
            m.conv2.weight, m.conv2.bias,
            m.conv3.weight, m.conv3.bias,
        ]
    weights += [
        m.fc.weight, m.fc.bias,
        m.policy.weight, m.policy.bias,
        m.baseline.weight, m.baseline.bias,
    ]
    num_layers = 0
    if m.use_lstm:
        num_layers = m.core.num_layers
        for layer in range(num_layers):
            weights += [
                getattr(m.core, f"weight_ih_l{layer}"),
                getattr(m.core, f"weight_hh_l{layer}"),
                getattr(m.core, f"bias_ih_l{layer}"),
                getattr(m.core, f"bias_hh_l{layer}"),
            ]
    return runtime._tbruntime.InferenceRunner(
        inference_batcher, [w.detach() for w in weights], num_layers, greedy,
        model_type,
    )

_inference_stream = None
_inference_stream_lock = threading.Lock()


def _get_inference_stream(device):
    """One shared side stream for ALL inference work, so behavior-model
    forwards never queue behind the learner's step on the default stream
    (the reference serializes both on one device context; on MI355X the
    side stream + pinned copies make inference latency independent of the
    learner)."""
    global _inference_stream
    if device.type != "cuda":
        return None
    with _inference_stream_lock:
        if _inference_stream is None:
            _inference_stream = torch.cuda.Stream(device=device)
    return _inference_stream


def inference(flags, inference_batcher, model, lock=threading.Lock()):  # noqa: B008
    """Consume inference batches: one batched behavior-model forward each."""
    stream = _get_inference_stream(flags.actor_device)
    import contextlib

    stream_ctx = (
        torch.cuda.stream(stream) if stream is not None else contextlib.nullcontext()
    )
    with torch.no_grad(), stream_ctx:
        for batch in inference_batcher:
            INFERENCE_STATS["batches"] += 1
            INFERENCE_STATS["env_steps"] += batch.size()
            batched_env_outputs, agent_state = batch.get_inputs()
            frame, reward, done, *_ = batched_env_outputs
            frame = frame.to(flags.actor_device, non_blocking=True)
            reward = reward.to(flags.actor_device, non_blocking=True)
            done = done.to(flags.actor_device, non_blocking=True)
            agent_state = tuple(
                t.to(flags.actor_device, non_blocking=True) for t in agent_state
            )
            with lock:
                outputs, new_state = model(
                    dict(frame=frame, reward=reward, done=done), agent_state
                )
                outputs = _as_agent_output(outputs)
                outputs = tuple(t.cpu() for t in outputs)
                new_state = tuple(t.cpu() for t in new_state)
            batch.set_outputs((outputs, new_state))


def learn(
    flags,
    learner_queue,
    model,
    flat_param,
    flat_grad,
    actor_flat,
    optimizer,
    scheduler,
    stats,
    plogger,
    reducer,
    num_updates,
    update_counter,
    inference_runner=None,
    lock=threading.Lock(),  # noqa: B008
):
    """Learner loop: exactly num_updates optimizer steps (shared across
    threads via update_counter), identical on every DP rank so collectives
    stay aligned."""

    def device_step(env_outputs, actor_outputs, initial_agent_state):
        """Everything between batch-on-GPU and host-side bookkeeping: model
        forward, V-trace, losses, backward, all-reduce, fused optimizer,
        behavior-model sync. Pure device work so it is hipGraph-capturable."""
        learner_outputs, _ = model(
            dict(
                frame=env_outputs.frame,
                reward=env_outputs.rewards,
                done=env_outputs.done,
            ),
            initial_agent_state,
        )
        learner_outputs = AgentOutput._make(_as_agent_output(learner_outputs))

        bootstrap_value = learner_outputs.baseline[-1]

        # Shift: env_outputs[t+1] is the consequence of actions[t].
        env_outputs = EnvOutput._make(t[1:] for t in env_outputs)
        actor_outputs = AgentOutput._make(t[1:] for t in actor_outputs)
        learner_outputs = AgentOutput._make(t[:-1] for t in learner_outputs)

        if flags.reward_clipping == "abs_one":
            clipped_rewards = torch.clamp(env_outputs.rewards, -1, 1)
        else:
            clipped_rewards = env_outputs.rewards

        discounts = (~env_outputs.done).float() * flags.discounting

        vtrace_returns = vtrace.from_logits(
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
            vtrace_returns.pg_advantages,
            vtrace_returns.vs,
        )
        total_loss = (
            pg_loss
            + flags.baseline_cost * baseline_loss
            + flags.entropy_cost * entropy_loss
        )

        optimizer.zero_grad()
        total_loss.backward()
        reducer.reduce()
        optimizer.step()

        # Behavior-model sync: one flat copy (D2D on GPU).
        with torch.no_grad():
            actor_flat.copy_(flat_param)
        return total_loss, pg_loss, baseline_loss, entropy_loss

    use_graph = (
        getattr(flags, "use_hipgraph", False)
        and flags.learner_device.type == "cuda"
        and reducer.world_size == 1
        and flags.num_learner_threads == 1
        and not flags.use_lstm  # shallow AND deep capture (bench.py default);
        # the persistent LSTM gains nothing from replay (PROFILE_r2.md)
    )
    graph = None
    static_in = None
    static_out = None

    for tensors in learner_queue:
        tensors = tuple(
            t.to(flags.learner_device, non_blocking=True)
            for t in runtime._tbruntime.flatten(tensors)
        )
        # Structure: ((env_outputs[5], actor_outputs[3]), initial_agent_state).
        env_outputs = EnvOutput._make(tensors[:5])
        actor_outputs = AgentOutput._make(tensors[5:8])
        initial_agent_state = tensors[8:]

        with lock:
            with update_counter["mutex"]:
                if update_counter["done"] >= num_updates:
                    return
                update_counter["done"] += 1

            if use_graph and graph is None:
                # First batch on this thread: capture the device step.
                optimizer.enable_device_lr()
                static_in = (
                    EnvOutput._make(t.clone() for t in env_outputs),
                    AgentOutput._make(t.clone() for t in actor_outputs),
                    tuple(t.clone() for t in initial_agent_state),
                )
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    for _ in range(3):
                        device_step(*static_in)
                torch.cuda.current_stream().wait_stream(side)
                try:
                    graph = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(
                        graph, capture_error_mode="thread_local"
                    ):
                        static_out = device_step(*static_in)
                except Exception:
                    logging.exception(
                        "hipGraph capture failed; continuing eager")
                    use_graph = False
                    graph = None
                    torch.cuda.synchronize()

            if graph is not None:
                for dst, src in zip(
                    list(static_in[0]) + list(static_in[1])
                    + list(static_in[2]),
                    list(env_outputs) + list(actor_outputs)
                    + list(initial_agent_state),
                ):
                    dst.copy_(src, non_blocking=True)
                optimizer.push_lr()
                graph.replay()
                total_loss, pg_loss, baseline_loss, entropy_loss = static_out
                env_outputs = EnvOutput._make(
                    t[1:] for t in static_in[0])
            else:
                (total_loss, pg_loss, baseline_loss,
                 entropy_loss) = device_step(env_outputs, actor_outputs,
                                             initial_agent_state)
                env_outputs = EnvOutput._make(t[1:] for t in env_outputs)
            scheduler.step()
            if inference_runner is not None:
                inference_runner.mark_weights_dirty()

            episode_returns = env_outputs.episode_return[env_outputs.done]
            stats["step"] = (stats.get("step", 0)
                         + flags.unroll_length * flags.batch_size
                         * max(1, reducer.world_size))
            stats["episode_returns"] = tuple(episode_returns.cpu().numpy())
            stats["mean_episode_return"] = torch.mean(episode_returns).item()
            stats["mean_episode_step"] = torch.mean(
                env_outputs.episode_step.float()
            ).item()
            stats["total_loss"] = total_loss.item()
            stats["pg_loss"] = pg_loss.item()
            stats["baseline_loss"] = baseline_loss.item()
            stats["entropy_loss"] = entropy_loss.item()
            stats["learner_queue_size"] = learner_queue.size()

            if plogger is not None:
                plogger.log(stats)

            if not len(episode_returns):
                stats["mean_episode_return"] = None


def train(flags):  # noqa: C901
    rank, world_size, local_rank = tbddp.maybe_init_distributed()
    is_leader = rank == 0
    torch.manual_seed(4242 + rank * 977)

    if flags.xpid is None:
        flags.xpid = "polybeast-%s" % time.strftime("%Y%m%d-%H%M%S")
    plogger = None
    if is_leader:
        plogger = file_writer.FileWriter(
            xpid=flags.xpid, xp_args=flags.__dict__, rootdir=flags.savedir
        )
    checkpointpath = os.path.expandvars(
        os.path.expanduser("%s/%s/%s" % (flags.savedir, flags.xpid, "model.tar"))
    )

    if not flags.disable_cuda and torch.cuda.is_available():
        flags.learner_device = torch.device("cuda", local_rank)
        flags.actor_device = torch.device("cuda", local_rank)
        torch.cuda.set_device(local_rank)
    else:
        flags.learner_device = torch.device("cpu")
        flags.actor_device = torch.device("cpu")

    if flags.max_learner_queue_size is None:
        flags.max_learner_queue_size = flags.batch_size

    # Rollouts are batched along dim 1 ([T+1, B, ...]).
    learner_queue = runtime.BatchingQueue(
        batch_dim=1,
        minimum_batch_size=flags.batch_size,
        maximum_batch_size=flags.batch_size,
        check_inputs=True,
        maximum_queue_size=flags.max_learner_queue_size,
        # Batches are assembled straight into HBM (async DMA per rollout).
        output_device=(
            str(flags.learner_device)
            if flags.learner_device.type == "cuda"
            else None
        ),
    )
    inference_batcher = runtime.DynamicBatcher(
        batch_dim=1,
        minimum_batch_size=flags.inference_min_batch_size,
        maximum_batch_size=flags.inference_max_batch_size,
        timeout_ms=flags.inference_timeout_ms,
        check_outputs=True,
    )

    if tbflags.parse_synthetic_env_spec(flags.env) is not None:
        # In-process native synthetic envs; give each rank distinct streams.
        addresses = [
            f"{flags.env}" for _ in range(flags.num_actors)
        ]
    else:
        addresses = []
        basename = flags.pipes_basename
        for i in range(flags.num_actors):
            addresses.append(f"{basename}.{rank * flags.num_actors + i}")

    model = create_model(flags).to(device=flags.learner_device)
    actor_model = create_model(flags).to(device=flags.actor_device)

    flat_param = tbflat.flatten_parameters(model)
    flat_grad = tbflat.attach_flat_grads(model)
    actor_flat = tbflat.flatten_parameters(actor_model)

    # All ranks start from rank 0's init.
    tbddp.broadcast_flat(flat_param)
    with torch.no_grad():
        actor_flat.copy_(flat_param)

    if getattr(flags, "momentum", 0):
        raise ValueError(
            "polybeast's fused RMSProp implements momentum=0 semantics only; "
            "got --momentum %r (use monobeast for momentum>0)" % flags.momentum
        )
    optimizer = tbflat.FusedRMSProp(
        flat_param,
        flat_grad,
        lr=flags.learning_rate,
        alpha=flags.alpha,
        eps=flags.epsilon,
        clip_norm=flags.grad_norm_clipping,
    )
    steps_per_update = flags.unroll_length * flags.batch_size * world_size
    scheduler = tbflat.LinearLR(optimizer, steps_per_update, flags.total_steps)
    reducer = tbddp.GradAllReducer(
        flat_grad, world_size,
        params=[p for p in model.parameters() if p.requires_grad])

    stats = {}

    if getattr(flags, "checkpoint_exists_ok", True) and os.path.exists(
        checkpointpath
    ):
        logging.info("Resuming from checkpoint %s", checkpointpath)
        ckpt = torch.load(checkpointpath, map_location=flags.learner_device,
                          weights_only=False)
        model.load_state_dict(ckpt["model_state_dict"])
        optimizer.load_state_dict(ckpt["optimizer_state_dict"])
        scheduler.load_state_dict(ckpt["scheduler_state_dict"])
        stats = ckpt.get("stats", {})
        with torch.no_grad():
            actor_flat.copy_(flat_param)

    initial_agent_state = model.initial_state(batch_size=1)
    initial_agent_state = tuple(t.cpu() for t in initial_agent_state)

    use_cpp_inference = (
        flags.actor_device.type == "cuda"
        and not getattr(flags, "py_inference", False)
    )
    _shape, _ = observation_shape(flags)
    _frame_bytes = 1
    for _d in _shape:
        _frame_bytes *= _d
    use_obs_slab = use_cpp_inference and not os.environ.get(
        "TBAMD_NO_OBS_SLAB") and (
        bool(os.environ.get("TBAMD_OBS_SLAB")) or _frame_bytes > 48 * 1024)
    actor_pool = runtime.ActorPool(
        unroll_length=flags.unroll_length,
        learner_queue=learner_queue,
        inference_batcher=inference_batcher,
        env_server_addresses=addresses,
        initial_agent_state=initial_agent_state,
        seed_base=rank * flags.num_actors,
        # Actors publish observations into a pinned slab and requests carry
        # only slot ids (GPU-side gather). Default: on for large frames
        # (full-res per-batch pinned cat measured 26+ ms), off for 84x84
        # (pinned-cat + SDMA measured ~12% faster there).
        use_obs_slab=use_obs_slab,
        rollout_budget_mb=flags.rollout_buffer_budget_mb,
        envs_per_thread=(
            flags.envs_per_thread if flags.envs_per_thread > 0
            else (16 if tbflags.parse_synthetic_env_spec(flags.env) is not None
                  else 1)),
    )

    pool_failure = []

    def run_pool():
        try:
            actor_pool.run()
        except Exception as e:
            # Actor death must be LOUD: without this, a dropped first batch
            # (e.g. a serve error) kills every actor thread silently and
            # train() starves forever on an empty learner queue.
            logging.exception("Exception in actor pool")
            pool_failure.append(e)
            try:
                inference_batcher.close()
            except Exception:
                pass
            try:
                learner_queue.close()
            except Exception:
                pass

    actorpool_thread = threading.Thread(target=run_pool, name="actorpool")

    num_updates = max(1, -(-flags.total_steps // steps_per_update))  # ceil
    done_so_far = stats.get("step", 0) // (flags.unroll_length * flags.batch_size)
    update_counter = {"mutex": threading.Lock(), "done": done_so_far}

    inference_runner = None
    if use_cpp_inference:
        inference_runner = make_inference_runner(actor_model, inference_batcher)
        inference_threads = []
    learner_threads = [
        threading.Thread(
            target=learn,
            name=f"learner-{i}",
            args=(flags, learner_queue, model, flat_param, flat_grad, actor_flat,
                  optimizer, scheduler, stats, plogger, reducer, num_updates,
                  update_counter, inference_runner),
        )
        # Collectives must stay ordered: one learner thread under DP.
        for i in range(1 if world_size > 1 else flags.num_learner_threads)
    ]
    if not use_cpp_inference:
        inference_threads = [
            threading.Thread(
                target=inference,
                name=f"inference-{i}",
                args=(flags, inference_batcher, actor_model),
            )
            for i in range(flags.num_inference_threads)
        ]

    actorpool_thread.start()
    if inference_runner is not None:
        if use_obs_slab:
            slab = actor_pool.obs_slab()  # blocks until the first env obs
            if slab:
                inference_runner.set_obs_slab(*slab)
        inference_runner.start(flags.num_inference_threads)
    for t in learner_threads + inference_threads:
        t.start()

    def checkpoint():
        if flags.disable_checkpoint or not is_leader:
            return
        logging.info("Saving checkpoint to %s", checkpointpath)
        torch.save(
            {
                "model_state_dict": model.state_dict(),
                "optimizer_state_dict": optimizer.state_dict(),
                "scheduler_state_dict": scheduler.state_dict(),
                "stats": stats,
                "flags": vars(flags),
            },
            checkpointpath,
        )

    def format_value(x):
        return f"{x:1.5}" if isinstance(x, float) else str(x)

    timer = timeit.default_timer
    try:
        last_checkpoint_time = timer()
        while update_counter["done"] < num_updates and any(
            t.is_alive() for t in learner_threads
        ) and not pool_failure:
            start_time = timer()
            start_step = stats.get("step", 0)
            time.sleep(5)
            end_step = stats.get("step", 0)

            if timer() - last_checkpoint_time > 10 * 60:
                checkpoint()
                last_checkpoint_time = timer()

            if is_leader:
                logging.info(
                    "Step %i @ %.1f SPS (x%d ranks). Inference batcher size: %i."
                    " Learner queue size: %i. Other stats: (%s)",
                    end_step,
                    (end_step - start_step) / (timer() - start_time),
                    world_size,
                    inference_batcher.size(),
                    learner_queue.size(),
                    ", ".join(
                        f"{key} = {format_value(value)}"
                        for key, value in stats.items()
                        if key not in ("episode_returns",)
                    ),
                )
    except KeyboardInterrupt:
        pass
    finally:
        inference_batcher.close()
        learner_queue.close()
        if inference_runner is not None:
            inference_runner.stop()
        for t in learner_threads + inference_threads:
            t.join(timeout=10)
        actorpool_thread.join(timeout=10)
        checkpoint()
        if plogger is not None:
            plogger.close()

    if pool_failure:
        raise RuntimeError("actor pool failed during training") from pool_failure[0]
    logging.info("Rank %d done after %d updates.", rank, update_counter["done"])
    return model


def test(flags, num_episodes: int = 10):
    """Greedy-policy evaluation on a local env (the reference leaves
    polybeast test unimplemented, ref: polybeast_learner.py:596-597)."""
    from torchbeast_amd import monobeast
    from torchbeast_amd.core import environment

    if flags.xpid is None:
        checkpointpath = "./latest/model.tar"
    else:
        checkpointpath = os.path.expandvars(
            os.path.expanduser(f"{flags.savedir}/{flags.xpid}/model.tar")
        )

    model = create_model(flags)
    model.eval()
    checkpoint = torch.load(checkpointpath, map_location="cpu",
                            weights_only=False)
    model.load_state_dict(checkpoint["model_state_dict"])

    env = environment.Environment(monobeast.create_env(flags))
    observation = env.initial()
    core_state = model.initial_state(batch_size=1)
    returns = []
    while len(returns) < num_episodes:
        outputs, core_state = model(observation, core_state)
        action = _as_agent_output(outputs)[0]
        observation = env.step(action)
        if observation["done"].item():
            returns.append(observation["episode_return"].item())
            logging.info(
                "Episode ended after %d steps. Return: %.1f",
                observation["episode_step"].item(),
                observation["episode_return"].item(),
            )
    env.close()
    logging.info("Average returns over %i episodes: %.1f", num_episodes,
                 sum(returns) / len(returns))
    return sum(returns) / len(returns)


def main(flags):
    if not hasattr(flags, "checkpoint_exists_ok"):
        flags.checkpoint_exists_ok = True
    if flags.write_profiler_trace:
        logging.info("Running with profiler.")
        with torch.profiler.profile(
            activities=[
                torch.profiler.ProfilerActivity.CPU,
                torch.profiler.ProfilerActivity.CUDA,
            ]
        ) as prof:
            if flags.mode == "train":
                train(flags)
            else:
                test(flags)
        filename = "chrome-%s.trace" % time.strftime("%Y%m%d-%H%M%S")
        logging.info("Writing profiler trace to '%s'", filename)
        prof.export_chrome_trace(filename)
    else:
        if flags.mode == "train":
            train(flags)
        else:
            test(flags)


if __name__ == "__main__":
    main(parser.parse_args())
