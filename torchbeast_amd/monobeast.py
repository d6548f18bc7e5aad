"""MonoBeast: single-machine IMPALA (ref: torchbeast/monobeast.py).

Actors are forked processes, each stepping its own environment and a shared
CPU copy of the model; rollouts travel through shared-memory tensor buffers
indexed by free/full queues; learner threads stack buffers into [T+1, B]
batches, move them to the GPU, and run V-trace + losses + RMSProp.

MI355X specifics relative to the reference:
- On ROCm devices the V-trace scan, the fused loss and the LSTM unroll
  dispatch to CDNA4 HIP kernels (torchbeast_amd/ops/).
- Batches are staged through pinned host memory so the H2D copy is a DMA on
  a side stream (`--pin_buffers`).

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

# Re-exported for API parity with the reference module.
Net = AtariNet

logging.basicConfig(
    format="[%(levelname)s:%(process)d %(module)s:%(lineno)d %(asctime)s] %(message)s",
    level=0,
)

Buffers = typing.Dict[str, typing.List[torch.Tensor]]


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


def act(
    flags,
    actor_index: int,
    free_queue: mp.SimpleQueue,
    full_queue: mp.SimpleQueue,
    model: torch.nn.Module,
    buffers: Buffers,
    initial_agent_state_buffers,
):
    """Actor process: step the env with the shared behavior model, writing
    each unroll into the shared-memory buffer slot it takes from free_queue."""
    try:
        # Actors are forked: cap intra-op threads BEFORE the first tensor op
        # (a multi-threaded OpenMP pool inherited across fork deadlocks), and
        # one thread per actor is the right sizing anyway.
        torch.set_num_threads(1)
        logging.info("Actor %i started.", actor_index)
        timings = prof.Timings()

        gym_env = create_env(flags)
        if hasattr(gym_env, "seed"):
            seed = actor_index ^ int.from_bytes(os.urandom(4), byteorder="little")
            gym_env.seed(seed)
        env = environment.Environment(gym_env)
        env_output = env.initial()
        agent_state = model.initial_state(batch_size=1)
        agent_output, unused_state = model(env_output, agent_state)

        while True:
            index = free_queue.get()
            if index is None:
                break

            # Step 0 of the new rollout is the final step of the previous one.
            for key in env_output:
                buffers[key][index][0, ...] = env_output[key]
            for key in agent_output:
                buffers[key][index][0, ...] = agent_output[key]
            for i, t in enumerate(agent_state):
                initial_agent_state_buffers[index][i][...] = t

            for t in range(flags.unroll_length):
                timings.reset()

                with torch.no_grad():
                    agent_output, agent_state = model(env_output, agent_state)
                timings.time("model")

                env_output = env.step(agent_output["action"])
                timings.time("step")

                for key in env_output:
                    buffers[key][index][t + 1, ...] = env_output[key]
                for key in agent_output:
                    buffers[key][index][t + 1, ...] = agent_output[key]
                timings.time("write")

            full_queue.put(index)

        if actor_index == 0:
            logging.info("Actor 0 timings: %s", timings.summary())

    except KeyboardInterrupt:
        pass  # Silently exit on ctrl-c; the main process handles shutdown.
    except Exception:
        logging.error("Exception in actor %i:\n%s", actor_index, traceback.format_exc())
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
    with lock:
        timings.time("lock")
        indices = [full_queue.get() for _ in range(flags.batch_size)]
        timings.time("dequeue")
    batch = {
        key: torch.stack([buffers[key][m] for m in indices], dim=1) for key in buffers
    }
    initial_agent_state = (
        torch.cat(ts, dim=1)
        for ts in zip(*[initial_agent_state_buffers[m] for m in indices])
    )
    timings.time("batch")
    for m in indices:
        free_queue.put(m)
    timings.time("enqueue")
    batch = {k: t.to(device=flags.device, non_blocking=True) for k, t in batch.items()}
    initial_agent_state = tuple(
        t.to(device=flags.device, non_blocking=True) for t in initial_agent_state
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
    """One learner step: forward, V-trace targets, losses, backward, RMSProp,
    and a weight push back to the shared behavior model."""
    with lock:
        learner_outputs, unused_state = model(batch, initial_agent_state)

        # Bootstrap from the value estimate at the final step.
        bootstrap_value = learner_outputs["baseline"][-1]

        # Align: env_output[t+1] is the consequence of agent_output[t].
        batch = {key: tensor[1:] for key, tensor in batch.items()}
        learner_outputs = {key: tensor[:-1] for key, tensor in learner_outputs.items()}

        rewards = batch["reward"]
        if flags.reward_clipping == "abs_one":
            clipped_rewards = torch.clamp(rewards, -1, 1)
        else:
            clipped_rewards = rewards

        discounts = (~batch["done"]).float() * flags.discounting

        vtrace_returns = vtrace.from_logits(
            behavior_policy_logits=batch["policy_logits"],
            target_policy_logits=learner_outputs["policy_logits"],
            actions=batch["action"],
            discounts=discounts,
            rewards=clipped_rewards,
            values=learner_outputs["baseline"],
            bootstrap_value=bootstrap_value,
        )

        pg_loss, baseline_loss, entropy_loss = tbops.fused_impala_loss(
            learner_outputs["policy_logits"],
            learner_outputs["baseline"],
            batch["action"],
            vtrace_returns.pg_advantages,
            vtrace_returns.vs,
        )
        total_loss = (
            pg_loss
            + flags.baseline_cost * baseline_loss
            + flags.entropy_cost * entropy_loss
        )

        episode_returns = batch["episode_return"][batch["done"]]
        stats = {
            "episode_returns": tuple(episode_returns.tolist()),
            "mean_episode_return": torch.mean(episode_returns).item(),
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


def create_buffers(flags, obs_shape, num_actions) -> Buffers:
    T = flags.unroll_length
    specs = dict(
        frame=dict(size=(T + 1, *obs_shape), dtype=torch.uint8),
        reward=dict(size=(T + 1,), dtype=torch.float32),
        done=dict(size=(T + 1,), dtype=torch.bool),
        episode_return=dict(size=(T + 1,), dtype=torch.float32),
        episode_step=dict(size=(T + 1,), dtype=torch.int32),
        policy_logits=dict(size=(T + 1, num_actions), dtype=torch.float32),
        baseline=dict(size=(T + 1,), dtype=torch.float32),
        last_action=dict(size=(T + 1,), dtype=torch.int64),
        action=dict(size=(T + 1,), dtype=torch.int64),
    )
    buffers: Buffers = {key: [] for key in specs}
    for _ in range(flags.num_buffers):
        for key in buffers:
            buffers[key].append(torch.empty(**specs[key]).share_memory_())
    return buffers


def train(flags):  # noqa: C901
    if flags.xpid is None:
        flags.xpid = "torchbeast-%s" % time.strftime("%Y%m%d-%H%M%S")
    plogger = file_writer.FileWriter(
        xpid=flags.xpid, xp_args=flags.__dict__, rootdir=flags.savedir
    )
    checkpointpath = os.path.expandvars(
        os.path.expanduser("%s/%s/%s" % (flags.savedir, flags.xpid, "model.tar"))
    )

    if flags.num_buffers is None:
        flags.num_buffers = max(2 * flags.num_actors, flags.batch_size)
    if flags.num_actors >= flags.num_buffers:
        raise ValueError("num_buffers should be larger than num_actors")
    if flags.num_buffers < flags.batch_size:
        raise ValueError("num_buffers should be larger than batch_size")

    T = flags.unroll_length
    B = flags.batch_size

    flags.device = None
    if not flags.disable_cuda and torch.cuda.is_available():
        logging.info("Using CUDA (ROCm).")
        flags.device = torch.device("cuda")
    else:
        logging.info("Not using CUDA.")
        flags.device = torch.device("cpu")

    env = create_env(flags)
    obs_shape = env.reset().shape if hasattr(env, "reset") else env.observation_space.shape
    num_actions = (
        env.num_actions
        if hasattr(env, "num_actions")
        else env.action_space.n
    )

    model = Net(obs_shape, num_actions, flags.use_lstm)
    buffers = create_buffers(flags, obs_shape, model.num_actions)

    model.share_memory()

    # Shared slots for the recurrent state at each rollout's first step.
    initial_agent_state_buffers = []
    for _ in range(flags.num_buffers):
        state = model.initial_state(batch_size=1)
        for t in state:
            t.share_memory_()
        initial_agent_state_buffers.append(state)

    actor_processes = []
    ctx = mp.get_context("fork")
    free_queue = ctx.SimpleQueue()
    full_queue = ctx.SimpleQueue()

    for i in range(flags.num_actors):
        actor = ctx.Process(
            target=act,
            args=(
                flags,
                i,
                free_queue,
                full_queue,
                model,
                buffers,
                initial_agent_state_buffers,
            ),
        )
        actor.start()
        actor_processes.append(actor)

    learner_model = Net(obs_shape, num_actions, flags.use_lstm).to(
        device=flags.device
    )
    learner_model.load_state_dict(model.state_dict())

    optimizer = torch.optim.RMSprop(
        learner_model.parameters(),
        lr=flags.learning_rate,
        momentum=flags.momentum,
        eps=flags.epsilon,
        alpha=flags.alpha,
    )

    def lr_lambda(epoch):
        return 1 - min(epoch * T * B, flags.total_steps) / flags.total_steps

    scheduler = torch.optim.lr_scheduler.LambdaLR(optimizer, lr_lambda)

    logger = logging.getLogger("logfile")
    stat_keys = [
        "total_loss",
        "mean_episode_return",
        "pg_loss",
        "baseline_loss",
        "entropy_loss",
    ]
    logger.info("# Step\t%s", "\t".join(stat_keys))

    step, stats = 0, {}

    def batch_and_learn(i, lock=threading.Lock()):
        """Learner thread: repeatedly assemble a batch and run learn()."""
        nonlocal step, stats
        timings = prof.Timings()
        while step < flags.total_steps:
            timings.reset()
            batch, agent_state = get_batch(
                flags,
                free_queue,
                full_queue,
                buffers,
                initial_agent_state_buffers,
                timings,
            )
            stats = learn(
                flags, model, learner_model, batch, agent_state, optimizer, scheduler
            )
            timings.time("learn")
            with lock:
                to_log = dict(step=step)
                to_log.update({k: stats[k] for k in stat_keys})
                plogger.log(to_log)
                step += T * B

        if i == 0:
            logging.info("Batch and learn: %s", timings.summary())

    for m in range(flags.num_buffers):
        free_queue.put(m)

    threads = []
    for i in range(flags.num_learner_threads):
        thread = threading.Thread(
            target=batch_and_learn, name="batch-and-learn-%d" % i, args=(i,)
        )
        thread.start()
        threads.append(thread)

    def checkpoint():
        if flags.disable_checkpoint:
            return
        logging.info("Saving checkpoint to %s", checkpointpath)
        torch.save(
            {
                "model_state_dict": learner_model.state_dict(),
                "optimizer_state_dict": optimizer.state_dict(),
                "scheduler_state_dict": scheduler.state_dict(),
                "flags": vars(flags),
            },
            checkpointpath,
        )

    timer = timeit.default_timer
    try:
        last_checkpoint_time = timer()
        while step < flags.total_steps:
            start_step = step
            start_time = timer()
            time.sleep(5)

            if timer() - last_checkpoint_time > 10 * 60:
                checkpoint()
                last_checkpoint_time = timer()

            sps = (step - start_step) / (timer() - start_time)
            if stats.get("episode_returns", None):
                mean_return = (
                    "Return per episode: %.1f. " % stats["mean_episode_return"]
                )
            else:
                mean_return = ""
            total_loss = stats.get("total_loss", float("inf"))
            logging.info(
                "Steps %i @ %.1f SPS. Loss %f. %sStats:\n%s",
                step,
                sps,
                total_loss,
                mean_return,
                pprint.pformat(stats),
            )
    except KeyboardInterrupt:
        return  # Try joining actors then quit.
    else:
        for thread in threads:
            thread.join()
        logging.info("Learning finished after %d steps.", step)
    finally:
        for _ in range(flags.num_actors):
            free_queue.put(None)
        for actor in actor_processes:
            actor.join(timeout=1)
        checkpoint()
        plogger.close()


def test(flags, num_episodes: int = 10):
    if flags.xpid is None:
        checkpointpath = "./latest/model.tar"
    else:
        checkpointpath = os.path.expandvars(
            os.path.expanduser("%s/%s/%s" % (flags.savedir, flags.xpid, "model.tar"))
        )

    gym_env = create_env(flags)
    env = environment.Environment(gym_env)
    obs_shape = gym_env.reset().shape if hasattr(gym_env, "reset") else gym_env.observation_space.shape
    num_actions = (
        gym_env.num_actions
        if hasattr(gym_env, "num_actions")
        else gym_env.action_space.n
    )
    model = Net(obs_shape, num_actions, flags.use_lstm)
    model.eval()
    checkpoint = torch.load(checkpointpath, map_location="cpu", weights_only=False)
    model.load_state_dict(checkpoint["model_state_dict"])

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
    logging.info(
        "Average returns over %i episodes: %.1f", num_episodes, sum(returns) / len(returns)
    )


def main(flags):
    if flags.mode == "train":
        train(flags)
    else:
        test(flags)


if __name__ == "__main__":
    main(parser.parse_args())
