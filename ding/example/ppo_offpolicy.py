"""CartPole off-policy PPO pipeline (reference
ding/example/ppo_offpolicy.py)."""
from ding.data import DequeBuffer
from ding.framework import OnlineRLContext, task
from ding.framework.middleware import (
    OffPolicyLearner, StepCollector, gae_estimator, interaction_evaluator, termination_checker,
)
from ding.policy import PPOOffPolicy
from ding.envs import BaseEnvManagerV2
from ding.utils import EasyDict
from .common import cartpole_envs, compile


def main(max_step: int = 100, exp_name: str = 'exp/example_ppo_offpolicy'):
    main_config = EasyDict(dict(
        exp_name=exp_name,
        env=dict(collector_env_num=4, evaluator_env_num=4, n_evaluator_episode=4, stop_value=195),
        policy=dict(
            cuda=False, action_space='discrete',
            model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[64, 64]),
            learn=dict(update_per_collect=2, batch_size=64, learning_rate=3e-4, epoch_per_collect=1),
            collect=dict(n_sample=128, discount_factor=0.99, gae_lambda=0.95, unroll_len=1),
            eval=dict(evaluator=dict(eval_freq=100)),
            other=dict(replay_buffer=dict(replay_buffer_size=4096)),
        ),
    ))
    create_config = EasyDict(dict(
        env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
        env_manager=dict(type='base'),
        policy=dict(type='ppo_offpolicy'),
    ))
    cfg = compile(main_config, create_config, exp_name)
    ce, ee = cartpole_envs(cfg)
    policy = PPOOffPolicy(cfg.policy)
    buffer_ = DequeBuffer(size=cfg.policy.other.replay_buffer.replay_buffer_size)
    with task.start(ctx=OnlineRLContext()):
        task.use(interaction_evaluator(cfg, policy.eval_mode, ee))
        task.use(StepCollector(cfg, policy.collect_mode, ce))
        task.use(gae_estimator(cfg, policy.collect_mode, buffer_))
        task.use(OffPolicyLearner(cfg, policy.learn_mode, buffer_))
        task.use(termination_checker(max_env_step=int(1e5)))
        task.run(max_step=max_step)
    ce.close()
    ee.close()
    return policy


if __name__ == '__main__':
    main()
