"""torchrun driver: run the ACTUAL polybeast_learner.train() under
data-parallel (gloo on CPU here; the identical code path is RCCL on
MI355X), then dump this rank's post-training flat parameters and step
count so the test can assert replica identity across ranks."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from torchbeast_amd import polybeast_learner
from torchbeast_amd.parallel import flat as tbflat


def main():
    outdir = sys.argv[1]
    rank = int(os.environ.get("RANK", "0"))
    flags = polybeast_learner.parser.parse_args([])
    flags.env = "synthetic:4x36x36:6"
    flags.savedir = outdir
    flags.xpid = "ddptrain"
    flags.num_actors = 2
    flags.batch_size = 2
    flags.unroll_length = 8
    flags.total_steps = 256
    flags.num_learner_threads = 1
    flags.num_inference_threads = 1
    flags.disable_cuda = True

    model = polybeast_learner.train(flags)
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    torch.save({"flat": flat}, os.path.join(outdir, f"rank{rank}.pt"))


if __name__ == "__main__":
    main()
