"""High-level one-class-per-algorithm Agent API.

Parity: reference ding/bonus (DQNAgent/PPOF/SACAgent/... with
train/deploy/collect_data/batch_evaluate, bonus/__init__.py:13-25,
ppof.py:199-403). Agents bind an env id + algorithm default config and drive
the serial pipelines under the hood.
"""
import copy
import os
from dataclasses import dataclass
from typing import Any, List, Optional, Union

import torch

from ding.config import compile_config
from ding.entry import serial_pipeline, serial_pipeline_onpolicy, collect_demo_data, eval as eval_entry
from ding.policy import create_policy
from ding.utils import EasyDict, deep_merge_dicts


@dataclass
class TrainingReturn:
    wandb_url: Optional[str] = None
    stop_value: float = 0.0
    converged: bool = False


@dataclass
class EvalReturn:
    eval_value: float = 0.0
    eval_value_std: float = 0.0


# built-in env presets (offline-capable envs from this repo's dizoo)
ENV_PRESETS = {
    'CartPole-v0': dict(
        env=dict(
            type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env'],
            collector_env_num=4, evaluator_env_num=4, n_evaluator_episode=4, stop_value=195,
        ),
        obs_shape=4,
        action_shape=2,
        action_space='discrete',
    ),
    'Pendulum-v1': dict(
        env=dict(
            type='pendulum', import_names=['dizoo.classic_control.pendulum.envs.pendulum_env'],
            collector_env_num=4, evaluator_env_num=4, n_evaluator_episode=4, stop_value=-250, act_scale=True,
        ),
        obs_shape=3,
        action_shape=1,
        action_space='continuous',
    ),
}


class BaseAgent:
    """Shared train/deploy/collect_data/batch_evaluate plumbing."""

    policy_type: str = None
    example_algo: str = None  # ding/config/example/<dir> with tuned presets
    on_policy_pipeline: bool = False
    default_policy_cfg: dict = {}

    def __init__(
        self,
        env_id: str = 'CartPole-v0',
        exp_name: Optional[str] = None,
        seed: int = 0,
        cfg: Optional[dict] = None,
        policy_state_dict: Optional[Union[str, dict]] = None,
    ):
        self.env_id = env_id
        self.seed = seed
        self.exp_name = exp_name or f"{env_id}-{self.policy_type}"
        example = None
        if self.example_algo is not None:
            from ding.config.example import get_example_config
            example = get_example_config(self.example_algo, env_id)
        if example is not None:
            # tuned per-env preset from ding/config/example/<ALGO>/
            main = example
            main.exp_name = self.exp_name
        else:
            assert env_id in ENV_PRESETS, f"unknown env preset {env_id}; available: {list(ENV_PRESETS)}"
            preset = copy.deepcopy(ENV_PRESETS[env_id])
            main = EasyDict(dict(
                exp_name=self.exp_name,
                env=preset['env'],
                policy=deep_merge_dicts(
                    dict(model=dict(obs_shape=preset['obs_shape'], action_shape=preset['action_shape'])),
                    copy.deepcopy(self.default_policy_cfg)
                ),
            ))
        if cfg:
            main = EasyDict(deep_merge_dicts(main, cfg))
        create = EasyDict(dict(
            env=dict(type=main.env['type'], import_names=main.env['import_names']),
            env_manager=dict(type='base'),
            policy=dict(type=self.policy_type),
        ))
        self.main_config = main
        self.create_config = create
        self._policy = None
        self._pending_state_dict = policy_state_dict

    # -------------------------------------------------------------- train
    def train(self, step: int = int(1e7), max_train_iter: int = int(1e9), collector_env_num: Optional[int] = None,
              evaluator_env_num: Optional[int] = None, debug: bool = False, **kwargs) -> TrainingReturn:
        main = copy.deepcopy(self.main_config)
        if collector_env_num:
            main.env.collector_env_num = collector_env_num
        if evaluator_env_num:
            main.env.evaluator_env_num = evaluator_env_num
        pipeline = serial_pipeline_onpolicy if self.on_policy_pipeline else serial_pipeline
        self._policy = pipeline(
            (main, copy.deepcopy(self.create_config)), seed=self.seed, max_env_step=step,
            max_train_iter=max_train_iter
        )
        return TrainingReturn(stop_value=self.main_config.env.stop_value)

    def _ensure_policy(self):
        if self._policy is None:
            import copy as _c
            from ding.entry.serial_entry import _prepare
            cfg, policy, ce, ee = _prepare(
                (_c.deepcopy(self.main_config), _c.deepcopy(self.create_config)), self.seed, None, None
            )
            ce.close()
            ee.close()
            self._policy = policy
            if self._pending_state_dict is not None:
                sd = self._pending_state_dict
                if isinstance(sd, str):
                    sd = torch.load(sd, map_location='cpu', weights_only=False)
                policy.learn_mode.load_state_dict(sd)
        return self._policy

    # ------------------------------------------------------------- deploy
    def deploy(self, enable_save_replay: bool = False, replay_save_path: Optional[str] = None,
               seed: Optional[int] = None, debug: bool = False, max_episode_steps: int = 10000) -> EvalReturn:
        """Run one episode with the current policy in a fresh env."""
        from ding.envs import create_env
        policy = self._ensure_policy()
        env_cfg = EasyDict(copy.deepcopy(self.main_config.env))
        env_cfg.type = self.create_config.env.type
        env_cfg.import_names = self.create_config.env.import_names
        env = create_env(env_cfg)
        env.seed(seed if seed is not None else self.seed, dynamic_seed=False)
        obs = env.reset()
        total = 0.0
        fwd = policy.eval_mode.forward
        for _ in range(max_episode_steps):
            out = fwd({0: torch.as_tensor(obs, dtype=torch.float32)})
            action = out[0]['action'].numpy()
            ts = env.step(action)
            total += float(torch.as_tensor(ts.reward).reshape(-1)[0])
            obs = ts.obs
            if ts.done:
                break
        env.close()
        return EvalReturn(eval_value=total)

    # ------------------------------------------------------- collect/eval
    def collect_data(self, env_num: int = 4, save_data_path: Optional[str] = None, n_sample: int = 1000,
                     context=None, debug: bool = False) -> None:
        policy = self._ensure_policy()
        save_data_path = save_data_path or os.path.join(self.exp_name, 'demo_data.pkl')
        collect_demo_data(
            (copy.deepcopy(self.main_config), copy.deepcopy(self.create_config)), self.seed, n_sample,
            expert_data_path=save_data_path,
            state_dict=policy.collect_mode.state_dict(),
        )

    def batch_evaluate(self, env_num: int = 4, n_evaluator_episode: int = 4, context=None,
                       debug: bool = False) -> EvalReturn:
        policy = self._ensure_policy()
        value = eval_entry(
            (copy.deepcopy(self.main_config), copy.deepcopy(self.create_config)), self.seed,
            state_dict=policy.eval_mode.state_dict(),
        )
        return EvalReturn(eval_value=value)

    @property
    def best(self) -> 'BaseAgent':
        ckpt = os.path.join(self.exp_name, 'ckpt', 'ckpt_best.pth.tar')
        if os.path.exists(ckpt):
            self._pending_state_dict = ckpt
            self._policy = None
        return self


class DQNAgent(BaseAgent):
    policy_type = 'dqn'
    example_algo = 'DQN'
    default_policy_cfg = dict(
        nstep=3, discount_factor=0.97,
        learn=dict(update_per_collect=5, batch_size=64, learning_rate=1e-3),
        collect=dict(n_sample=32),
        other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
                   replay_buffer=dict(replay_buffer_size=20000)),
    )


class PPOF(BaseAgent):
    """PPO-of-fans: the simplified high-level PPO (on-policy pipeline)."""
    policy_type = 'ppo'
    example_algo = 'PPOF'
    on_policy_pipeline = True
    default_policy_cfg = dict(
        action_space='discrete',
        model=dict(action_space='discrete'),
        learn=dict(epoch_per_collect=4, batch_size=64, learning_rate=3e-4),
        collect=dict(n_sample=256, discount_factor=0.99, gae_lambda=0.95),
    )


class PPOOffPolicyAgent(BaseAgent):
    policy_type = 'ppo_offpolicy'
    example_algo = 'PPOOffPolicy'
    default_policy_cfg = dict(
        model=dict(action_space='discrete'),
        learn=dict(update_per_collect=4, batch_size=64, epoch_per_collect=1),
        collect=dict(n_sample=128),
        other=dict(replay_buffer=dict(replay_buffer_size=10000)),
    )


class A2CAgent(BaseAgent):
    policy_type = 'a2c'
    example_algo = 'A2C'
    on_policy_pipeline = True
    default_policy_cfg = dict(
        learn=dict(batch_size=64, learning_rate=1e-3),
        collect=dict(n_sample=64),
    )


class C51Agent(BaseAgent):
    policy_type = 'c51'
    example_algo = 'C51'
    default_policy_cfg = dict(
        nstep=3,
        model=dict(v_min=-10, v_max=10, n_atom=51),
        learn=dict(update_per_collect=5, batch_size=64, learning_rate=1e-3),
        collect=dict(n_sample=32),
        other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
                   replay_buffer=dict(replay_buffer_size=20000)),
    )


class SACAgent(BaseAgent):
    policy_type = 'sac'
    example_algo = 'SAC'
    default_policy_cfg = dict(
        random_collect_size=100,
        model=dict(action_space='reparameterization', twin_critic=True),
        learn=dict(update_per_collect=2, batch_size=64),
        collect=dict(n_sample=32),
        other=dict(replay_buffer=dict(replay_buffer_size=100000)),
    )


class DDPGAgent(BaseAgent):
    policy_type = 'ddpg'
    example_algo = 'DDPG'
    default_policy_cfg = dict(
        random_collect_size=100,
        model=dict(action_space='regression'),
        learn=dict(update_per_collect=2, batch_size=64, learning_rate_actor=1e-3, learning_rate_critic=1e-3),
        collect=dict(n_sample=32),
        other=dict(replay_buffer=dict(replay_buffer_size=100000)),
    )


class TD3Agent(BaseAgent):
    policy_type = 'td3'
    example_algo = 'TD3'
    default_policy_cfg = DDPGAgent.default_policy_cfg


class SQLAgent(BaseAgent):
    policy_type = 'sql'
    example_algo = 'SQL'
    default_policy_cfg = DQNAgent.default_policy_cfg


class PGAgent(BaseAgent):
    policy_type = 'pg'
    example_algo = 'PG'
    on_policy_pipeline = True
    default_policy_cfg = dict(
        learn=dict(batch_size=64, learning_rate=1e-3),
        collect=dict(n_sample=64, collector=dict(type='episode', get_train_sample=True)),
    )
