"""PolyBeast environment servers (ref: torchbeast/polybeast_env.py).

Spawns `--num_servers` processes, each hosting environments behind the
unix-socket step protocol at `{pipes_basename}.{i}`. Built-in env kinds:

- "Mock" / "synthetic[:CxHxW[:A]]": gym-free synthetic Atari-shaped frames
  (the benchmark envs; note the learner can also run these in-process
  without any server — see polybeast_learner).
- any gym Atari id (requires gym): full deepmind preprocessing stack.
"""

import argparse
import logging
import multiprocessing as mp
import time

from torchbeast_amd import flags as tbflags
from torchbeast_amd import runtime

logging.basicConfig(
    format="[%(levelname)s:%(process)d %(module)s:%(lineno)d %(asctime)s] %(message)s",
    level=0,
)

parser = argparse.ArgumentParser(description="MI355X-native PolyBeast environments")
parser.add_argument("--pipes_basename", default="unix:/tmp/polybeast",
                    help="Basename for the env-server unix sockets.")
parser.add_argument("--num_servers", default=4, type=int,
                    help="Number of env server processes.")
parser.add_argument("--env", type=str, default="Mock",
                    help="Env name: 'Mock', 'synthetic[:CxHxW[:A]]', or a gym id.")


def create_env_factory(env_name):
    if env_name == "Mock" or env_name.startswith("synthetic"):
        spec = tbflags.parse_synthetic_env_spec(
            env_name if env_name.startswith("synthetic") else "synthetic"
        )

        def factory():
            from torchbeast_amd.envs.synthetic import SyntheticAtariEnv

            return SyntheticAtariEnv(shape=spec[0], num_actions=spec[1])

        return factory

    def factory():
        from torchbeast_amd.envs import atari

        return atari.wrap_pytorch(
            atari.wrap_deepmind(
                atari.make_atari(env_name),
                clip_rewards=False,
                frame_stack=True,
                scale=False,
            )
        )

    return factory


def serve(env_name, address):
    factory = create_env_factory(env_name)
    server = runtime.Server(factory, address)
    logging.info("Starting env server on %s", address)
    server.run()


def main(flags):
    processes = []
    for i in range(flags.num_servers):
        p = mp.Process(
            target=serve,
            args=(flags.env, f"{flags.pipes_basename}.{i}"),
            daemon=True,
        )
        p.start()
        processes.append(p)

    try:
        # We are a process group; wait for the children.
        while True:
            time.sleep(10)
    except KeyboardInterrupt:
        pass


if __name__ == "__main__":
    main(parser.parse_args())
