"""MonoBeast: single-machine IMPALA (capability parity with
torchbeast/monobeast.py — fresh MI355X-native implementation).

Topology: forked actor processes each own an environment plus the shared
CPU behavior model and fill shared-memory rollout slots, handing slot ids
through free/full queues; learner threads stack slots into [T+1, B] batches,
push them to the GPU and run V-trace + IMPALA losses + RMSProp, then copy
fresh weights back into the shared behavior model.

MI355X specifics: on ROCm devices the V-trace scan, the fused loss, the
conv trunk and the LSTM unroll dispatch to the CDNA4 HIP kernels in
torchbeast_amd/ops/.

Run: python -m torchbeast_amd.monobeast --env synthetic --num_actors 4
"""

import argparse
import logging
import os
import pprint
import threading
import time
import timeit
import traceback
import typing

import torch
from torch import multiprocessing as mp
from torch import nn

from torchbeast_amd import flags as tbflags
from torchbeast_amd.core import environment, file_writer, prof, vtrace
from torchbeast_amd.models.atari_net import AtariNet
from torchbeast_amd.ops import functional as tbops

# The reference exports its model under both names; keep that surface.
Net = AtariNet

logging.basicConfig(
    format="[%(levelname)s:%(process)d %(module)s:%(lineno)d %(asctime)s] %(message)s",
    level=0,
)

Buffers = typing.Dict[str, typing.List[torch.Tensor]]

CHECKPOINT_INTERVAL_S = 10 * 60
MONITOR_INTERVAL_S = 5


def make_parser():
    parser = argparse.ArgumentParser(description="MI355X-native MonoBeast")
    parser.add_argument("--mode", default="train",
                        choices=["train", "test", "test_render"])
    tbflags.add_common_flags(parser)
    parser.add_argument("--num_actors", default=4, type=int, metavar="N",
                        help="Number of actor processes.")
    parser.add_argument("--num_buffers", default=None, type=int, metavar="N",
                        help="Number of shared-memory rollout buffers.")
    parser.add_argument("--num_learner_threads", "--num_threads", default=2,
                        type=int, metavar="N", help="Number of learner threads.")
    parser.add_argument("--pin_buffers", action="store_true",
                        help="Host-register rollout buffers for DMA H2D copies.")
    tbflags.add_loss_flags(parser)
    tbflags.add_optimizer_flags(parser)
    return parser


parser = make_parser()


def create_env(flags):
    spec = tbflags.parse_synthetic_env_spec(flags.env)
    if spec is not None:
        from torchbeast_amd.envs.synthetic import SyntheticAtariEnv

        return SyntheticAtariEnv(shape=spec[0], num_actions=spec[1])
    from torchbeast_amd.envs import atari

    return atari.wrap_pytorch(
        atari.wrap_deepmind(
            atari.make_atari(flags.env),
            clip_rewards=False,
            frame_stack=True,
            scale=False,
        )
    )


def _env_interface(env):
    """(obs_shape, num_actions) for either gym-style or synthetic envs."""
    if hasattr(env, "num_actions"):
        return env.reset().shape, env.num_actions
    return env.observation_space.shape, env.action_space.n


def _checkpoint_path(flags):
    return os.path.expandvars(
        os.path.expanduser(f"{flags.savedir}/{flags.xpid}/model.tar")
    )


def create_buffers(flags, obs_shape, num_actions) -> Buffers:
    """Shared-memory rollout slots: one [T+1, ...] tensor per field per slot."""
    T1 = flags.unroll_length + 1
    field_specs = [
        ("frame", (T1, *obs_shape), torch.uint8),
        ("reward", (T1,), torch.float32),
        ("done", (T1,), torch.bool),
        ("episode_return", (T1,), torch.float32),
        ("episode_step", (T1,), torch.int32),
        ("policy_logits", (T1, num_actions), torch.float32),
        ("baseline", (T1,), torch.float32),
        ("last_action", (T1,), torch.int64),
        ("action", (T1,), torch.int64),
    ]
    buffers: Buffers = {}
    for name, shape, dtype in field_specs:
        buffers[name] = [
            torch.empty(shape, dtype=dtype).share_memory_()
            for _ in range(flags.num_buffers)
        ]
    return buffers


def act(
    flags,
    actor_index: int,
    free_queue: mp.SimpleQueue,
    full_queue: mp.SimpleQueue,
    model: torch.nn.Module,
    buffers: Buffers,
    initial_agent_state_buffers,
):
    """Actor-process main: run the env with the shared behavior model and
    fill whichever rollout slot free_queue hands out."""
    try:
        # We are a fork: pin intra-op threads to 1 BEFORE any tensor op (an
        # inherited multi-threaded OpenMP pool deadlocks across fork), and
        # a single thread per actor is the right sizing anyway.
        torch.set_num_threads(1)
        logging.info("Actor %d up.", actor_index)
        timings = prof.Timings()

        raw_env = create_env(flags)
        if hasattr(raw_env, "seed"):
            raw_env.seed(actor_index)
        env = environment.Environment(raw_env)

        step_out = env.initial()
        agent_state = model.initial_state(batch_size=1)
        agent_out, _ = model(step_out, agent_state)

        def record(slot, t):
            for name, value in step_out.items():
                buffers[name][slot][t, ...] = value
            for name, value in agent_out.items():
                buffers[name][slot][t, ...] = value

        while (slot := free_queue.get()) is not None:
            # Rollouts overlap by one step: row 0 repeats the previous
            # rollout's final step; the recurrent state snapshot goes with it.
            record(slot, 0)
            for i, s in enumerate(agent_state):
                initial_agent_state_buffers[slot][i][...] = s

            for t in range(flags.unroll_length):
                timings.reset()
                with torch.no_grad():
                    agent_out, agent_state = model(step_out, agent_state)
                timings.time("model")
                step_out = env.step(agent_out["action"])
                timings.time("step")
                record(slot, t + 1)
                timings.time("write")

            full_queue.put(slot)

        if actor_index == 0:
            logging.info("Actor 0 timings: %s", timings.summary())
    except KeyboardInterrupt:
        pass  # Parent coordinates shutdown.
    except Exception:
        logging.error("Actor %d died:\n%s", actor_index, traceback.format_exc())
        raise


def get_batch(
    flags,
    free_queue: mp.SimpleQueue,
    full_queue: mp.SimpleQueue,
    buffers: Buffers,
    initial_agent_state_buffers,
    timings,
    lock=threading.Lock(),
):
    """Claim batch_size full slots, stack them along dim 1, release the
    slots, and move everything to the learner device."""
    with lock:
        timings.time("lock")
        slots = [full_queue.get() for _ in range(flags.batch_size)]
        timings.time("dequeue")

    batch = {
        name: torch.stack([column[s] for s in slots], dim=1)
        for name, column in buffers.items()
    }
    state_columns = zip(*(initial_agent_state_buffers[s] for s in slots))
    initial_agent_state = [torch.cat(col, dim=1) for col in state_columns]
    timings.time("batch")

    for s in slots:
        free_queue.put(s)
    timings.time("enqueue")

    batch = {
        name: t.to(device=flags.device, non_blocking=True)
        for name, t in batch.items()
    }
    initial_agent_state = tuple(
        t.to(device=flags.device, non_blocking=True)
        for t in initial_agent_state
    )
    timings.time("device")
    return batch, initial_agent_state


def learn(
    flags,
    actor_model,
    model,
    batch,
    initial_agent_state,
    optimizer,
    scheduler,
    lock=threading.Lock(),
):
    """One optimizer step on a [T+1, B] batch, then a weight push to the
    shared behavior model."""
    with lock:
        learner_outputs, _ = model(batch, initial_agent_state)

        # The value estimate at the last row bootstraps the returns.
        bootstrap_value = learner_outputs["baseline"][-1]

        # Re-align so that row t pairs action[t] with its consequence.
        batch = {name: t[1:] for name, t in batch.items()}
        learner_outputs = {name: t[:-1] for name, t in learner_outputs.items()}

        if flags.reward_clipping == "abs_one":
            rewards = torch.clamp(batch["reward"], -1, 1)
        else:
            rewards = batch["reward"]
        discounts = (~batch["done"]).float() * flags.discounting

        targets = vtrace.from_logits(
            behavior_policy_logits=batch["policy_logits"],
            target_policy_logits=learner_outputs["policy_logits"],
            actions=batch["action"],
            discounts=discounts,
            rewards=rewards,
            values=learner_outputs["baseline"],
            bootstrap_value=bootstrap_value,
        )

        pg_loss, baseline_loss, entropy_loss = tbops.fused_impala_loss(
            learner_outputs["policy_logits"],
            learner_outputs["baseline"],
            batch["action"],
            targets.pg_advantages,
            targets.vs,
        )
        total_loss = (
            pg_loss
            + flags.baseline_cost * baseline_loss
            + flags.entropy_cost * entropy_loss
        )

        finished = batch["episode_return"][batch["done"]]
        stats = {
            "episode_returns": tuple(finished.tolist()),
            "mean_episode_return": torch.mean(finished).item(),
            "total_loss": total_loss.item(),
            "pg_loss": pg_loss.item(),
            "baseline_loss": baseline_loss.item(),
            "entropy_loss": entropy_loss.item(),
        }

        optimizer.zero_grad()
        total_loss.backward()
        nn.utils.clip_grad_norm_(model.parameters(), flags.grad_norm_clipping)
        optimizer.step()
        scheduler.step()

        actor_model.load_state_dict(model.state_dict())
        return stats


def train(flags):  # noqa: C901
    if flags.xpid is None:
        flags.xpid = time.strftime("torchbeast-%Y%m%d-%H%M%S")
    plogger = file_writer.FileWriter(
        xpid=flags.xpid, xp_args=flags.__dict__, rootdir=flags.savedir
    )

    if flags.num_buffers is None:
        flags.num_buffers = max(2 * flags.num_actors, flags.batch_size)
    if flags.num_actors >= flags.num_buffers:
        raise ValueError("num_buffers should be larger than num_actors")
    if flags.num_buffers < flags.batch_size:
        raise ValueError("num_buffers should be larger than batch_size")

    T, B = flags.unroll_length, flags.batch_size
    steps_per_update = T * B

    use_gpu = not flags.disable_cuda and torch.cuda.is_available()
    flags.device = torch.device("cuda" if use_gpu else "cpu")
    logging.info("Learner device: %s", flags.device)

    probe_env = create_env(flags)
    obs_shape, num_actions = _env_interface(probe_env)

    model = Net(obs_shape, num_actions, flags.use_lstm)
    model.share_memory()
    buffers = create_buffers(flags, obs_shape, model.num_actions)

    # One shared recurrent-state snapshot per rollout slot.
    initial_agent_state_buffers = []
    for _ in range(flags.num_buffers):
        snapshot = model.initial_state(batch_size=1)
        for s in snapshot:
            s.share_memory_()
        initial_agent_state_buffers.append(snapshot)

    ctx = mp.get_context("fork")
    free_queue = ctx.SimpleQueue()
    full_queue = ctx.SimpleQueue()

    actor_processes = []
    for i in range(flags.num_actors):
        p = ctx.Process(
            target=act,
            args=(flags, i, free_queue, full_queue, model, buffers,
                  initial_agent_state_buffers),
        )
        p.start()
        actor_processes.append(p)

    if flags.pin_buffers and flags.device.type == "cuda":
        # Page-lock the already-shared rollout slots in this (learner)
        # process so get_batch's H2D copies are true async DMA. hipHostRegister
        # works on fork-shared mappings; the actors keep writing through their
        # own (unregistered) mapping of the same pages.
        cudart = torch.cuda.cudart()
        for tensors in buffers.values():
            for t in tensors:
                cudart.cudaHostRegister(t.data_ptr(), t.numel() * t.element_size(), 0)

    learner_model = Net(obs_shape, num_actions, flags.use_lstm)
    learner_model = learner_model.to(device=flags.device)
    learner_model.load_state_dict(model.state_dict())

    optimizer = torch.optim.RMSprop(
        learner_model.parameters(),
        lr=flags.learning_rate,
        momentum=flags.momentum,
        eps=flags.epsilon,
        alpha=flags.alpha,
    )
    # Linear decay to zero over total_steps (in env steps).
    scheduler = torch.optim.lr_scheduler.LambdaLR(
        optimizer,
        lambda n_updates: 1
        - min(n_updates * steps_per_update, flags.total_steps)
        / flags.total_steps,
    )

    stat_keys = ["total_loss", "mean_episode_return", "pg_loss",
                 "baseline_loss", "entropy_loss"]
    logging.getLogger("logfile").info("# Step\t%s", "\t".join(stat_keys))

    step, stats = 0, {}

    def learner_thread_main(thread_id, log_lock=threading.Lock()):
        nonlocal step, stats
        timings = prof.Timings()
        while step < flags.total_steps:
            timings.reset()
            batch, agent_state = get_batch(
                flags, free_queue, full_queue, buffers,
                initial_agent_state_buffers, timings,
            )
            stats = learn(flags, model, learner_model, batch, agent_state,
                          optimizer, scheduler)
            timings.time("learn")
            with log_lock:
                row = {"step": step, **{k: stats[k] for k in stat_keys}}
                plogger.log(row)
                step += steps_per_update
        if thread_id == 0:
            logging.info("Learner timings: %s", timings.summary())

    for slot in range(flags.num_buffers):
        free_queue.put(slot)

    threads = [
        threading.Thread(target=learner_thread_main, args=(i,),
                         name=f"learner-{i}")
        for i in range(flags.num_learner_threads)
    ]
    for t in threads:
        t.start()

    def checkpoint():
        if flags.disable_checkpoint:
            return
        path = _checkpoint_path(flags)
        logging.info("Checkpointing to %s", path)
        torch.save(
            {
                "model_state_dict": learner_model.state_dict(),
                "optimizer_state_dict": optimizer.state_dict(),
                "scheduler_state_dict": scheduler.state_dict(),
                "flags": vars(flags),
            },
            path,
        )

    clock = timeit.default_timer
    try:
        last_ckpt = clock()
        while step < flags.total_steps:
            window_start_step, window_start_time = step, clock()
            time.sleep(MONITOR_INTERVAL_S)

            if clock() - last_ckpt > CHECKPOINT_INTERVAL_S:
                checkpoint()
                last_ckpt = clock()

            sps = (step - window_start_step) / (clock() - window_start_time)
            if stats.get("episode_returns"):
                ret_info = f"Return per episode: {stats['mean_episode_return']:.1f}. "
            else:
                ret_info = ""
            logging.info(
                "Steps %i @ %.1f SPS. Loss %f. %sStats:\n%s",
                step, sps, stats.get("total_loss", float("inf")), ret_info,
                pprint.pformat(stats),
            )
    except KeyboardInterrupt:
        return  # Fall through to finally for shutdown.
    else:
        for t in threads:
            t.join()
        logging.info("Learning finished after %d steps.", step)
    finally:
        for _ in range(flags.num_actors):
            free_queue.put(None)
        for p in actor_processes:
            p.join(timeout=1)
        checkpoint()
        plogger.close()


def test(flags, num_episodes: int = 10):
    if flags.xpid is None:
        checkpointpath = "./latest/model.tar"
    else:
        checkpointpath = _checkpoint_path(flags)

    raw_env = create_env(flags)
    obs_shape, num_actions = _env_interface(raw_env)
    env = environment.Environment(raw_env)

    model = Net(obs_shape, num_actions, flags.use_lstm)
    model.eval()  # Greedy action selection.
    snapshot = torch.load(checkpointpath, map_location="cpu",
                          weights_only=False)
    model.load_state_dict(snapshot["model_state_dict"])

    observation = env.initial()
    agent_state = model.initial_state(batch_size=1)
    returns = []
    while len(returns) < num_episodes:
        if flags.mode == "test_render":
            env.env.render()
        policy_outputs, agent_state = model(observation, agent_state)
        observation = env.step(policy_outputs["action"])
        if observation["done"].item():
            returns.append(observation["episode_return"].item())
            logging.info(
                "Episode ended after %d steps. Return: %.1f",
                observation["episode_step"].item(),
                observation["episode_return"].item(),
            )
    env.close()
    logging.info("Average returns over %i episodes: %.1f",
                 num_episodes, sum(returns) / len(returns))


def main(flags):
    if flags.mode == "train":
        train(flags)
    else:
        test(flags)


if __name__ == "__main__":
    main(parser.parse_args())
