"""Pendulum SAC off-policy pipeline (reference ding/example/sac.py)."""
from ding.data import DequeBuffer
from ding.framework import OnlineRLContext, task
from ding.framework.middleware import (
    OffPolicyLearner, StepCollector, data_pusher, interaction_evaluator, termination_checker,
)
from ding.policy import SACPolicy
from ding.utils import EasyDict
from .common import compile, pendulum_envs


def main(max_step: int = 300, exp_name: str = 'exp/example_sac'):
    main_config = EasyDict(dict(
        exp_name=exp_name,
        env=dict(collector_env_num=4, evaluator_env_num=4, n_evaluator_episode=4, stop_value=-250,
                 act_scale=True),
        policy=dict(
            cuda=False, random_collect_size=200,
            model=dict(obs_shape=3, action_shape=1, twin_critic=True, action_space='reparameterization'),
            learn=dict(update_per_collect=2, batch_size=64, auto_alpha=True),
            collect=dict(n_sample=32, unroll_len=1),
            eval=dict(evaluator=dict(eval_freq=100)),
            other=dict(replay_buffer=dict(replay_buffer_size=20000)),
        ),
    ))
    create_config = EasyDict(dict(
        env=dict(type='pendulum', import_names=['dizoo.classic_control.pendulum.envs.pendulum_env']),
        env_manager=dict(type='base'),
        policy=dict(type='sac'),
    ))
    cfg = compile(main_config, create_config, exp_name)
    collector_env, evaluator_env = pendulum_envs(cfg)
    policy = SACPolicy(cfg.policy)
    buffer_ = DequeBuffer(size=cfg.policy.other.replay_buffer.replay_buffer_size)
    with task.start(ctx=OnlineRLContext()):
        task.use(interaction_evaluator(cfg, policy.eval_mode, evaluator_env))
        task.use(StepCollector(cfg, policy.collect_mode, collector_env,
                               random_collect_size=cfg.policy.random_collect_size))
        task.use(data_pusher(cfg, buffer_))
        task.use(OffPolicyLearner(cfg, policy.learn_mode, buffer_))
        task.use(termination_checker(max_env_step=int(1e5)))
        task.run(max_step=max_step)
    collector_env.close()
    evaluator_env.close()
    return policy


if __name__ == '__main__':
    main()
