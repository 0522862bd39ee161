"""Sparse-reward gridworld DQN + RND intrinsic reward middleware
(reference ding/example/dqn_rnd.py)."""
from ding.data import DequeBuffer
from ding.framework import OnlineRLContext, task
from ding.framework.middleware import (
    OffPolicyLearner, StepCollector, data_pusher, eps_greedy_handler, interaction_evaluator,
    reward_estimator, termination_checker,
)
from ding.policy import DQNPolicy
from ding.reward_model import create_reward_model
from ding.envs import BaseEnvManagerV2
from ding.utils import EasyDict
from .common import compile


def main(max_step: int = 300, exp_name: str = 'exp/example_dqn_rnd'):
    from dizoo.gridworld.envs.minigrid_lite_env import MiniGridLiteEnv
    obs_dim = 5 * 5 * 4 + 4
    main_config = EasyDict(dict(
        exp_name=exp_name,
        env=dict(collector_env_num=2, evaluator_env_num=2, n_evaluator_episode=2, stop_value=2),
        policy=dict(
            cuda=False, nstep=1, discount_factor=0.99,
            model=dict(obs_shape=obs_dim, action_shape=3, encoder_hidden_size_list=[64, 64]),
            learn=dict(update_per_collect=2, batch_size=32, learning_rate=1e-3),
            collect=dict(n_sample=32),
            eval=dict(evaluator=dict(eval_freq=100)),
            other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
                       replay_buffer=dict(replay_buffer_size=5000)),
        ),
        reward_model=dict(type='rnd', obs_shape=obs_dim, hidden_size_list=[32, 32], update_per_collect=2),
    ))
    create_config = EasyDict(dict(
        env=dict(type='minigrid_lite', import_names=['dizoo.gridworld.envs.minigrid_lite_env']),
        env_manager=dict(type='base'),
        policy=dict(type='dqn'),
    ))
    cfg = compile(main_config, create_config, exp_name)
    ce = BaseEnvManagerV2(env_fn=[lambda: MiniGridLiteEnv({'grid_size': 5}) for _ in range(2)],
                          cfg=cfg.env.manager)
    ee = BaseEnvManagerV2(env_fn=[lambda: MiniGridLiteEnv({'grid_size': 5}) for _ in range(2)],
                          cfg=cfg.env.manager)
    ce.seed(0)
    ee.seed(0, dynamic_seed=False)
    policy = DQNPolicy(cfg.policy)
    buffer_ = DequeBuffer(size=cfg.policy.other.replay_buffer.replay_buffer_size)
    rnd = create_reward_model(cfg.reward_model, device='cpu')
    with task.start(ctx=OnlineRLContext()):
        task.use(interaction_evaluator(cfg, policy.eval_mode, ee))
        task.use(eps_greedy_handler(cfg))
        task.use(StepCollector(cfg, policy.collect_mode, ce))
        task.use(reward_estimator(cfg, rnd))
        task.use(data_pusher(cfg, buffer_))
        task.use(OffPolicyLearner(cfg, policy.learn_mode, buffer_))
        task.use(termination_checker(max_env_step=int(1e5)))
        task.run(max_step=max_step)
    ce.close()
    ee.close()
    return policy


if __name__ == '__main__':
    main()
