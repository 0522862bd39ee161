"""Self-play PPO training loop on GameEnv: two PPO policies collect battles
through BattleSampleSerialCollector, each learns from its own stream, and a
BattleInteractionSerialEvaluator pits policy 0 against a fixed opponent.

Parity: reference dizoo/league_demo/selfplay_demo_ppo_main.py (same loop
expressed with this build's worker classes).
"""
import numpy as np
import torch

from ding.config import compile_config
from ding.envs import create_env_manager
from ding.policy import PPOPolicy
from ding.utils import EasyDict, deep_merge_dicts, set_pkg_seed
from ding.worker import BattleSampleSerialCollector, BattleInteractionSerialEvaluator
from dizoo.league_demo.game_env import GameEnv
from dizoo.league_demo.selfplay_demo_ppo_config import selfplay_demo_ppo_config


class _FlattenPPO:
    """GameEnv emits a 2x2 payoff obs; flatten it for the MLP encoder."""

    def __init__(self, inner):
        self._inner = inner

    def forward(self, obs, **kwargs):
        flat = {i: torch.as_tensor(np.asarray(o), dtype=torch.float32).flatten() for i, o in obs.items()}
        return self._inner.forward(flat, **kwargs)

    def process_transition(self, obs, policy_output, timestep):
        flat_obs = torch.as_tensor(np.asarray(obs), dtype=torch.float32).flatten()
        flat_next = torch.as_tensor(np.asarray(timestep.obs), dtype=torch.float32).flatten()
        timestep = timestep._replace(obs=flat_next)
        return self._inner.process_transition(flat_obs, policy_output, timestep)

    def __getattr__(self, name):
        return getattr(self._inner, name)


class RandomEvalPolicy:

    def forward(self, data):
        return {eid: {'action': torch.randint(0, 2, (1, ))} for eid in data}

    def reset(self, data_id=None):
        pass


def main(cfg=None, seed: int = 0, max_train_iter: int = 100):
    cfg = EasyDict(cfg if cfg is not None else selfplay_demo_ppo_config)
    cfg = compile_config(cfg, seed=seed)
    set_pkg_seed(seed)
    env_cfg = {'game_type': cfg.env.env_type, 'repeat_count': cfg.env.repeat_count}
    collector_env = create_env_manager(
        EasyDict({'type': 'base'}), [lambda: GameEnv(dict(env_cfg)) for _ in range(cfg.env.collector_env_num)]
    )
    eval_env = create_env_manager(
        EasyDict({'type': 'base'}), [lambda: GameEnv(dict(env_cfg)) for _ in range(cfg.env.evaluator_env_num)]
    )
    pol_cfg = deep_merge_dicts(PPOPolicy.default_config(), cfg.policy)
    policies = [PPOPolicy(pol_cfg, enable_field=['learn', 'collect', 'eval']) for _ in range(2)]
    collector = BattleSampleSerialCollector(
        EasyDict({'type': 'sample_1v1'}), env=collector_env,
        policy=[_FlattenPPO(p.collect_mode) for p in policies], exp_name=cfg.exp_name
    )
    evaluator = BattleInteractionSerialEvaluator(
        EasyDict({'type': 'battle_interaction', 'eval_freq': 10, 'n_episode': cfg.env.n_evaluator_episode,
                  'stop_value': cfg.env.stop_value}),
        env=eval_env, policy=[_FlattenPPO(policies[0].eval_mode), RandomEvalPolicy()], exp_name=cfg.exp_name
    )
    for it in range(max_train_iter):
        if evaluator.should_eval(it):
            stop, info = evaluator.eval(None, train_iter=it, envstep=collector.envstep)
            if stop:
                break
        data, _ = collector.collect(n_sample=cfg.policy.collect.n_sample)
        for pid, policy in enumerate(policies):
            policy.learn_mode.forward(data[pid])
    collector.close()
    evaluator.close()
    return policies


if __name__ == '__main__':
    main()
