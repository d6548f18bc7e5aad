"""Inference-serving strategy comparison (analogue of the reference's
tests/inference_speed_profiling.py:38-99, which compared three CUDA
serving strategies with SPS + chrome traces).

Three MI355X strategies over the same DynamicBatcher workload:
  py-lock   : Python inference threads sharing the model under a lock
              (the reference's production path, polybeast_learner.py:269)
  cpp-aten  : the GIL-free C++ engine, library-op forward
              (TBAMD_TRUNK-style dispatch disabled -> at::conv2d path)
  cpp-fused : the C++ engine with the bf16 MFMA trunk + fused
              heads/sampling kernel (this repo's production path)

Run on a GPU box:
  python scripts/inference_speed_profiling.py [--requests 20000] [--trace]

--trace writes a chrome trace per strategy (torch.profiler, ROCm backend)
to ./inference_trace_<strategy>.json.gz.
"""

import argparse
import os
import sys
import threading
import timeit

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402

from torchbeast_amd import polybeast_learner as pbl  # noqa: E402
from torchbeast_amd import runtime  # noqa: E402
from torchbeast_amd.models.atari_net import AtariNet  # noqa: E402


def drive(batcher, n_requests, n_actors=256):
    """Feed the batcher with actor-shaped single-step requests from many
    threads; returns elapsed seconds for n_requests round trips."""
    per = n_requests // n_actors
    frame = torch.randint(0, 256, (1, 1, 4, 84, 84), dtype=torch.uint8)
    reward = torch.zeros(1, 1)
    done = torch.zeros(1, 1, dtype=torch.bool)
    fill = (frame, reward, done, torch.zeros(1, 1, dtype=torch.int32),
            torch.zeros(1, 1))

    errors = []

    def actor():
        try:
            for _ in range(per):
                batcher.compute((fill, ()))
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    threads = [threading.Thread(target=actor) for _ in range(n_actors)]
    t0 = timeit.default_timer()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    elapsed = timeit.default_timer() - t0
    if errors:
        raise errors[0]
    return elapsed, per * n_actors


def run_strategy(name, n_requests, trace):
    flags = pbl.parser.parse_args([])
    flags.actor_device = torch.device("cuda")
    model = AtariNet((4, 84, 84), 6, use_lstm=False,
                     use_last_action=False).cuda()
    batcher = runtime.DynamicBatcher(
        batch_dim=1, minimum_batch_size=64, maximum_batch_size=512,
        timeout_ms=5)

    stop = []
    runner = None
    py_threads = []
    if name == "py-lock":
        def py_loop():
            try:
                pbl.inference(flags, batcher, model)
            except runtime.ClosedBatchingQueue:
                pass

        py_threads = [threading.Thread(target=py_loop, daemon=True)
                      for _ in range(4)]
        for t in py_threads:
            t.start()
    else:
        if name == "cpp-aten":
            os.environ["TBAMD_RUNNER_ATEN"] = "1"
        else:
            os.environ.pop("TBAMD_RUNNER_ATEN", None)
        runner = pbl.make_inference_runner(model, batcher)
        runner.start(4)

    prof = None
    if trace:
        prof = torch.profiler.profile(
            activities=[torch.profiler.ProfilerActivity.CPU,
                        torch.profiler.ProfilerActivity.CUDA])
        prof.__enter__()
    # Warmup then timed.
    drive(batcher, 4096)
    elapsed, served = drive(batcher, n_requests)
    if prof is not None:
        prof.__exit__(None, None, None)
        path = f"inference_trace_{name}.json.gz"
        prof.export_chrome_trace(path)
        print(f"  trace -> {path}")

    batcher.close()
    if runner is not None:
        runner.stop()
    for t in py_threads:
        t.join(timeout=5)
    del stop
    return served / elapsed


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--requests", type=int, default=20000)
    p.add_argument("--trace", action="store_true")
    args = p.parse_args()
    torch.manual_seed(0)

    for name in ("cpp-fused", "cpp-aten", "py-lock"):
        sps = run_strategy(name, args.requests, args.trace)
        print(f"{name:>10}: {sps:,.0f} inference steps/s")


if __name__ == "__main__":
    main()
