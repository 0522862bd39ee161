"""Shortcut module: build a ding env from a gym env id. Parity: reference
ding/envs/gym_env.py. gym is optional in this image; the import error is
raised at call time, not import time."""
from ding.envs.env.ding_env_wrapper import DingEnvWrapper


def env(cfg, seed_api: bool = True, caller: str = 'collector', **kwargs):
    import gym
    return DingEnvWrapper(gym.make(cfg.env_id, **kwargs), cfg=cfg, seed_api=seed_api, caller=caller)
