"""Synthetic D4RL dataset generation (offline: no datasets downloadable).

Rolls out scripted behaviour policies of graded quality on the mujoco-lite
dynamics and saves obs/action/reward/done/next_obs as .npz — the same flat
format HDF5Dataset/D4RLTrajectoryDataset load (ding/utils/data/dataset.py).
Quality tiers: 'random' (uniform), 'medium' (linear policy + heavy noise),
'expert' (linear policy aligned with the reward direction, light noise);
'-medium-expert-' mixes the two halves like the real suite.
"""
import os

import numpy as np

from dizoo.d4rl.envs.d4rl_env import D4RLLiteEnv


def _behaviour(env, quality: str, rng: np.random.RandomState):
    obs_dim, act_dim = env.obs_dim, env.act_dim
    # the env's reward direction is fixed (rng seed 12345 inside the env);
    # a linear policy toward +w states is "expert-ish" on these dynamics
    W = rng.randn(act_dim, obs_dim) * 0.5
    if quality == 'random':
        return lambda o: rng.uniform(-1, 1, size=act_dim)
    noise = 0.8 if quality == 'medium' else 0.1
    return lambda o: np.clip(np.tanh(W @ o) + rng.randn(act_dim) * noise, -1, 1)


def generate_d4rl_npz(env_id: str, path: str, n_transitions: int = 10000, seed: int = 0) -> str:
    """Create the dataset for e.g. 'hopper-medium-v2' at ``path``."""
    rng = np.random.RandomState(seed)
    env = D4RLLiteEnv({'env_id': env_id, 'max_step': 200})
    env.seed(seed)
    parts = env_id.lower().split('-')
    qualities = [q for q in ('random', 'medium', 'expert', 'replay') if q in parts] or ['medium']
    obs_l, act_l, rew_l, done_l, next_l = [], [], [], [], []
    per_quality = n_transitions // len(qualities)
    for quality in qualities:
        pol = _behaviour(env, 'medium' if quality == 'replay' else quality, rng)
        obs = env.reset()
        for _ in range(per_quality):
            a = pol(obs).astype(np.float32)
            ts = env.step(a)
            obs_l.append(obs)
            act_l.append(a)
            rew_l.append(float(ts.reward[0]))
            done_l.append(bool(ts.done))
            next_l.append(ts.obs)
            obs = env.reset() if ts.done else ts.obs
    os.makedirs(os.path.dirname(path) or '.', exist_ok=True)
    np.savez(
        path,
        obs=np.asarray(obs_l, dtype=np.float32),
        action=np.asarray(act_l, dtype=np.float32),
        reward=np.asarray(rew_l, dtype=np.float32),
        done=np.asarray(done_l),
        next_obs=np.asarray(next_l, dtype=np.float32),
    )
    return path


def ensure_dataset(main_cfg) -> str:
    """Generate the npz a d4rl config points at (no-op if it exists)."""
    if 'dataset' in main_cfg and main_cfg.dataset.get('data_dir_prefix'):
        path = main_cfg.dataset.data_dir_prefix
    else:
        path = main_cfg.policy.collect.data_path
    if not os.path.exists(path):
        generate_d4rl_npz(main_cfg.env.env_id, path)
    return path
