"""Combined PolyBeast launcher (ref: torchbeast/polybeast.py): forks the env
server group, then runs the learner in this process."""

import argparse
import multiprocessing as mp

from torchbeast_amd import polybeast_env, polybeast_learner


def main():
    # Union of the learner and env flag namespaces via chained parsing.
    flags = argparse.Namespace()
    flags, argv = polybeast_learner.parser.parse_known_args(namespace=flags)
    flags, argv = polybeast_env.parser.parse_known_args(args=argv, namespace=flags)
    if argv:
        raise ValueError(f"Unkown args: {argv}")

    env_processes = []
    if not flags.env.startswith("synthetic"):
        # Synthetic envs run in-process in the ActorPool; everything else
        # gets a server process group.
        for i in range(flags.num_servers):
            p = mp.Process(
                target=polybeast_env.serve,
                args=(flags.env, f"{flags.pipes_basename}.{i}"),
                daemon=True,
            )
            p.start()
            env_processes.append(p)

    try:
        polybeast_learner.main(flags)
    finally:
        for p in env_processes:
            p.terminate()


if __name__ == "__main__":
    main()
