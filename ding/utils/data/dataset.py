"""Offline datasets.

Parity: reference ding/utils/data/dataset.py (NaiveRLDataset:30,
D4RLDataset:72, HDF5Dataset:259, D4RLTrajectoryDataset:411 for Decision
Transformer). Offline notes: no d4rl/h5py in the image — D4RL loading is
gated behind an informative error; HDF5Dataset also accepts .npz archives
with the same keys (obs/action/reward/done[/next_obs]).
"""
import os
import pickle
from typing import Any, Dict, List, Optional

import numpy as np
import torch
from torch.utils.data import Dataset

from ding.utils import DATASET_REGISTRY, EasyDict


@DATASET_REGISTRY.register('naive')
class NaiveRLDataset(Dataset):
    """A pickled list of transition dicts (from collect_demo_data)."""

    def __init__(self, cfg) -> None:
        if isinstance(cfg, str):
            self._data_path = cfg
        else:
            self._data_path = cfg.policy.collect.data_path
        with open(self._data_path, 'rb') as f:
            self._data: List[Dict[str, torch.Tensor]] = pickle.load(f)

    def __len__(self) -> int:
        return len(self._data)

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        return self._data[idx]


@DATASET_REGISTRY.register('hdf5')
class HDF5Dataset(Dataset):
    """obs/action/reward/done[/next_obs] arrays from .h5 (needs h5py) or .npz."""

    def __init__(self, cfg) -> None:
        data_path = cfg.policy.collect.data_path if not isinstance(cfg, str) else cfg
        norm_obs_cfg = None if isinstance(cfg, str) else cfg.policy.collect.get('normalize_states', None)
        if data_path.endswith('.npz'):
            archive = np.load(data_path)
            data = {k: archive[k] for k in archive.files}
        else:
            try:
                import h5py
            except ImportError:
                raise RuntimeError("h5py unavailable offline; convert the dataset to .npz")
            with h5py.File(data_path, 'r') as f:
                data = {k: f[k][()] for k in f.keys()}
        self._load(data, norm_obs_cfg)

    def _load(self, data: dict, norm_obs_cfg) -> None:
        self._data = []
        obs = data['obs'].astype(np.float32)
        if norm_obs_cfg:
            self.mean, self.std = obs.mean(0), obs.std(0) + 1e-3
            obs = (obs - self.mean) / self.std
        else:
            self.mean = self.std = None
        n = len(obs)
        next_obs = data.get('next_obs')
        if next_obs is None:
            next_obs = np.concatenate([obs[1:], obs[-1:]], axis=0)
        for i in range(n):
            self._data.append({
                'obs': torch.from_numpy(np.asarray(obs[i])),
                'action': torch.from_numpy(np.asarray(data['action'][i])),
                'reward': torch.tensor([float(np.asarray(data['reward'][i]).reshape(-1)[0])]),
                'done': bool(np.asarray(data['done'][i]).reshape(-1)[0]),
                'next_obs': torch.from_numpy(np.asarray(next_obs[i]).astype(np.float32)),
            })

    def __len__(self) -> int:
        return len(self._data)

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        return self._data[idx]


@DATASET_REGISTRY.register('d4rl')
class D4RLDataset(HDF5Dataset):

    def __init__(self, cfg) -> None:
        env_id = cfg.env.env_id if not isinstance(cfg, str) else cfg
        try:
            import gym
            import d4rl  # noqa
        except ImportError:
            raise RuntimeError(
                f"d4rl/gym unavailable offline: export '{env_id}' to .npz (obs/action/reward/done) "
                "and use dataset type 'hdf5'"
            )


@DATASET_REGISTRY.register('d4rl_trajectory')
class D4RLTrajectoryDataset(Dataset):
    """Context-window trajectory dataset for Decision Transformer.

    Source: pickled list of trajectories [{'observations': [T, obs],
    'actions': [T, act], 'rewards': [T]}], or .npz flat arrays + done splits.
    Returns (timesteps, states, actions, returns_to_go, traj_mask).
    """

    def __init__(self, cfg) -> None:
        if isinstance(cfg, str):
            data_path, context_len, rtg_scale = cfg, 20, 1000.0
        else:
            data_path = cfg.dataset.data_dir_prefix
            context_len = cfg.dataset.context_len
            rtg_scale = cfg.dataset.rtg_scale
        self.context_len = context_len
        if data_path.endswith('.npz'):
            archive = np.load(data_path)
            obs, act = archive['obs'], archive['action']
            rew, done = archive['reward'].reshape(-1), archive['done'].reshape(-1)
            self.trajectories = []
            start = 0
            for i in range(len(done)):
                if done[i] or i == len(done) - 1:
                    self.trajectories.append({
                        'observations': obs[start:i + 1],
                        'actions': act[start:i + 1],
                        'rewards': rew[start:i + 1],
                    })
                    start = i + 1
        else:
            with open(data_path, 'rb') as f:
                self.trajectories = pickle.load(f)
        states = np.concatenate([t['observations'] for t in self.trajectories], axis=0)
        self.state_mean, self.state_std = states.mean(0), states.std(0) + 1e-6
        for t in self.trajectories:
            # discounted=1 returns-to-go, normalized
            r = t['rewards']
            rtg = np.flip(np.cumsum(np.flip(r, 0)), 0).copy()
            t['returns_to_go'] = rtg / rtg_scale
            t['observations'] = (t['observations'] - self.state_mean) / self.state_std

    def get_state_stats(self):
        return self.state_mean, self.state_std

    def __len__(self) -> int:
        return len(self.trajectories)

    def __getitem__(self, idx: int):
        traj = self.trajectories[idx]
        T = len(traj['rewards'])
        C = self.context_len
        if T >= C:
            start = np.random.randint(0, T - C + 1)
            states = torch.from_numpy(traj['observations'][start:start + C]).float()
            actions = torch.from_numpy(traj['actions'][start:start + C]).float()
            rtg = torch.from_numpy(traj['returns_to_go'][start:start + C]).float().unsqueeze(-1)
            timesteps = torch.arange(start, start + C)
            mask = torch.ones(C, dtype=torch.long)
        else:
            pad = C - T
            states = torch.cat([torch.from_numpy(traj['observations']).float(),
                                torch.zeros(pad, *traj['observations'].shape[1:])], dim=0)
            actions = torch.cat([torch.from_numpy(traj['actions']).float(),
                                 torch.zeros(pad, *np.asarray(traj['actions']).shape[1:])], dim=0)
            rtg = torch.cat([torch.from_numpy(traj['returns_to_go']).float(),
                             torch.zeros(pad)], dim=0).unsqueeze(-1)
            timesteps = torch.arange(0, C)
            mask = torch.cat([torch.ones(T, dtype=torch.long), torch.zeros(pad, dtype=torch.long)])
        return timesteps, states, actions, rtg, mask


class DatasetStatistics:

    def __init__(self, mean, std, action_bounds=None):
        self.mean = mean
        self.std = std
        self.action_bounds = action_bounds


def create_dataset(cfg, **kwargs) -> Dataset:
    cfg = EasyDict(cfg)
    dataset_type = cfg.policy.collect.get('data_type', 'naive')
    return DATASET_REGISTRY.build(dataset_type, cfg=cfg, **kwargs)


def offline_data_save_type(exp_data: list, expert_data_path: str, data_type: str = 'naive') -> None:
    """Persist collected transitions for offline training."""
    d = os.path.dirname(expert_data_path)
    if d:
        os.makedirs(d, exist_ok=True)
    if data_type == 'naive':
        with open(expert_data_path, 'wb') as f:
            pickle.dump(exp_data, f)
    elif data_type in ('hdf5', 'npz'):
        arrays = {
            'obs': np.stack([np.asarray(t['obs']) for t in exp_data]),
            'action': np.stack([np.asarray(t['action']) for t in exp_data]),
            'reward': np.stack([np.asarray(t['reward']).reshape(-1)[0] for t in exp_data]),
            'done': np.stack([np.asarray(t['done']) for t in exp_data]),
            'next_obs': np.stack([np.asarray(t['next_obs']) for t in exp_data]),
        }
        path = expert_data_path if expert_data_path.endswith('.npz') else expert_data_path + '.npz'
        np.savez_compressed(path, **arrays)
    else:
        raise KeyError(data_type)


@DATASET_REGISTRY.register('bco')
class BCODataset(Dataset):
    """(obs, action) pair dataset for Behavioral Cloning from Observation —
    actions usually come from an inverse-dynamics model
    (ding.world_model.InverseDynamicsModel). Parity: reference
    ding/utils/data/dataset.py BCODataset:1136."""

    def __init__(self, data=None):
        if data is None:
            raise ValueError('Dataset can not be empty!')
        self._data = data

    def __len__(self):
        return len(self._data['obs'])

    def __getitem__(self, idx):
        return {k: v[idx] for k, v in self._data.items()}

    @property
    def obs(self):
        return self._data['obs']

    @property
    def action(self):
        return self._data['action']


def hdf5_save(exp_data, expert_data_path: str) -> None:
    """Save transitions as stacked arrays; uses the npz writer since h5py is
    not shipped in this image (reference dataset.py:1494)."""
    offline_data_save_type(exp_data, expert_data_path, data_type='hdf5')
