"""Serve your own environment to the polybeast learner.

The env plane speaks a framed tensor protocol over unix sockets or TCP
(runtime/csrc/wire.h); any object with the gym-style ``reset() -> obs``
and ``step(action) -> (obs, reward, done, info)`` methods can be served —
no gym dependency required. This mirrors the reference's custom-env flow
(torchbeast README "adding a new environment"), with `tcp:host:port`
addresses standing in for its gRPC channel.

Run the servers:

    python examples/custom_env.py --num_servers 4

then point the learner at them (same machine, unix sockets):

    python -m torchbeast_amd.polybeast_learner \
        --pipes_basename unix:/tmp/custom_env --num_actors 4 \
        --batch_size 4 --unroll_length 20 --total_steps 20000

For cross-machine serving pass e.g. ``--address tcp:0.0.0.0:7000`` here
and ``--pipes_basename tcp:envhost:7000`` to the learner.
"""

import argparse
import os
import sys
import threading

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from torchbeast_amd import runtime  # noqa: E402


class GridWorld:
    """Tiny deterministic grid: the agent walks a 1-D track of length 16;
    reward +1 at the right edge, episode ends after 64 steps. Observation
    is an Atari-shaped uint8 frame so the stock AtariNet accepts it."""

    TRACK = 16

    def __init__(self):
        self.pos = 0
        self.t = 0

    def _obs(self):
        frame = np.zeros((4, 84, 84), dtype=np.uint8)
        frame[:, :, self.pos * 84 // self.TRACK] = 255
        return frame

    def reset(self):
        self.pos, self.t = 0, 0
        return self._obs()

    def step(self, action):
        self.t += 1
        move = {0: 0, 1: 1, 2: -1}.get(int(action) % 3, 0)
        self.pos = min(self.TRACK - 1, max(0, self.pos + move))
        reward = 1.0 if self.pos == self.TRACK - 1 else 0.0
        done = self.t >= 64
        return self._obs(), reward, done, {}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--num_servers", type=int, default=4)
    p.add_argument("--address", default="unix:/tmp/custom_env",
                   help="Base address; server i appends '.{i}'. "
                        "Use tcp:host:port for cross-machine serving "
                        "(port is incremented per server).")
    args = p.parse_args()

    servers = []
    for i in range(args.num_servers):
        if args.address.startswith("tcp:"):
            _, host, port = args.address.split(":")
            addr = f"tcp:{host}:{int(port) + i}"
        else:
            addr = f"{args.address}.{i}"
        server = runtime.Server(GridWorld, addr)
        server.start()  # non-blocking; run() would block on the first
        servers.append(server)
        print(f"serving GridWorld on {addr}")

    try:
        threading.Event().wait()
    except KeyboardInterrupt:
        for s in servers:
            s.stop()


if __name__ == "__main__":
    main()
