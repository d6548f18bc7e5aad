from torchbeast_amd.envs.synthetic import CountingEnv, SyntheticAtariEnv

__all__ = ["SyntheticAtariEnv", "CountingEnv"]
