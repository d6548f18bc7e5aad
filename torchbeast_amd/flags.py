"""Shared argparse flag groups for the CLI entry points.

The reference repeats its argparse blocks per entry point
(ref: monobeast.py:40-94, polybeast_learner.py:38-102, polybeast_env.py:27-36);
here the groups are factored so monobeast/polybeast stay in sync. Flag names
and defaults match the reference exactly (unroll 80, batch 8, RMSProp
0.00048/0.99/0.01, entropy 0.0006, baseline 0.5, gamma 0.99, clip 40).
"""

import argparse


def add_common_flags(parser: argparse.ArgumentParser):
    parser.add_argument("--env", type=str, default="PongNoFrameskip-v4",
                        help="Gym environment name, or 'synthetic[:CxHxW[:A]]'.")
    parser.add_argument("--xpid", default=None, help="Experiment id.")
    parser.add_argument("--savedir", default="~/logs/torchbeast",
                        help="Root dir for experiment data.")
    parser.add_argument("--disable_checkpoint", action="store_true",
                        help="Disable checkpoint saving.")
    parser.add_argument("--use_lstm", action="store_true",
                        help="Use an LSTM core in the agent model.")
    parser.add_argument("--total_steps", default=100000, type=int, metavar="T",
                        help="Total environment steps to train for.")
    parser.add_argument("--batch_size", default=8, type=int, metavar="B",
                        help="Learner batch size.")
    parser.add_argument("--unroll_length", default=80, type=int, metavar="T",
                        help="Unroll length (time dimension).")
    parser.add_argument("--disable_cuda", action="store_true", help="Disable CUDA.")
    return parser


def add_loss_flags(parser: argparse.ArgumentParser):
    parser.add_argument("--entropy_cost", default=0.0006, type=float,
                        help="Entropy cost/multiplier.")
    parser.add_argument("--baseline_cost", default=0.5, type=float,
                        help="Baseline cost/multiplier.")
    parser.add_argument("--discounting", default=0.99, type=float,
                        help="Discount factor.")
    parser.add_argument("--reward_clipping", default="abs_one",
                        choices=["abs_one", "none"], help="Reward clipping.")
    return parser


def add_optimizer_flags(parser: argparse.ArgumentParser):
    parser.add_argument("--learning_rate", default=0.00048, type=float,
                        metavar="LR", help="Learning rate.")
    parser.add_argument("--alpha", default=0.99, type=float,
                        help="RMSProp smoothing constant.")
    parser.add_argument("--momentum", default=0, type=float,
                        help="RMSProp momentum.")
    parser.add_argument("--epsilon", default=0.01, type=float,
                        help="RMSProp epsilon.")
    parser.add_argument("--grad_norm_clipping", default=40.0, type=float,
                        help="Global gradient-norm clip.")
    return parser


def parse_synthetic_env_spec(env: str):
    """Parse 'synthetic[:CxHxW[:A]]' -> (shape, num_actions) or None."""
    if not env.startswith("synthetic"):
        return None
    shape, num_actions = (4, 84, 84), 6
    parts = env.split(":")
    if len(parts) >= 2 and parts[1]:
        shape = tuple(int(d) for d in parts[1].split("x"))
    if len(parts) >= 3 and parts[2]:
        num_actions = int(parts[2])
    return shape, num_actions
