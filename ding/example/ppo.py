"""CartPole on-policy PPO with GAE middleware (reference
ding/example/ppo.py)."""
from ding.framework import OnlineRLContext, task
from ding.framework.middleware import (
    StepCollector, gae_estimator, interaction_evaluator, multistep_trainer, termination_checker,
)
from ding.policy import PPOPolicy
from ding.utils import EasyDict
from .common import cartpole_envs, compile


def main(max_step: int = 1000, exp_name: str = 'exp/example_ppo'):
    main_config = EasyDict(dict(
        exp_name=exp_name,
        env=dict(collector_env_num=4, evaluator_env_num=4, n_evaluator_episode=4, stop_value=195),
        policy=dict(
            cuda=False, action_space='discrete', recompute_adv=True,
            model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[64, 64]),
            learn=dict(epoch_per_collect=2, batch_size=64, learning_rate=3e-4),
            collect=dict(n_sample=256, discount_factor=0.99, gae_lambda=0.95),
            eval=dict(evaluator=dict(eval_freq=100)),
        ),
    ))
    create_config = EasyDict(dict(
        env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
        env_manager=dict(type='base'),
        policy=dict(type='ppo'),
    ))
    cfg = compile(main_config, create_config, exp_name)
    collector_env, evaluator_env = cartpole_envs(cfg)
    policy = PPOPolicy(cfg.policy)
    with task.start(ctx=OnlineRLContext()):
        task.use(interaction_evaluator(cfg, policy.eval_mode, evaluator_env))
        task.use(StepCollector(cfg, policy.collect_mode, collector_env))
        task.use(gae_estimator(cfg, policy.collect_mode))
        task.use(multistep_trainer(policy.learn_mode))
        task.use(termination_checker(max_env_step=int(1e5)))
        task.run(max_step=max_step)
    collector_env.close()
    evaluator_env.close()
    return policy


if __name__ == '__main__':
    main()
