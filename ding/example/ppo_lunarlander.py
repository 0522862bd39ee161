"""LunarLander PPO middleware example (reference ding/example/ppo_lunarlander.py)."""
from ding.framework import OnlineRLContext, task
from ding.framework.middleware import (
    CkptSaver, StepCollector, gae_estimator, interaction_evaluator, multistep_trainer, termination_checker,
)
from ding.policy import PPOPolicy
from ding.utils import deep_merge_dicts
from .common import compile, lunarlander_envs


def main(max_step: int = 1000, exp_name: str = 'exp/example_ppo_lunarlander'):
    from dizoo.box2d.lunarlander.config.lunarlander_ppo_config import create_config, main_config
    cfg = compile(main_config, create_config, exp_name)
    collector_env, evaluator_env = lunarlander_envs(cfg)
    policy = PPOPolicy(deep_merge_dicts(PPOPolicy.default_config(), cfg.policy))
    with task.start(ctx=OnlineRLContext()):
        task.use(interaction_evaluator(cfg, policy.eval_mode, evaluator_env))
        task.use(StepCollector(cfg, policy.collect_mode, collector_env))
        task.use(gae_estimator(cfg, policy.collect_mode))
        task.use(multistep_trainer(policy.learn_mode))
        task.use(CkptSaver(policy, cfg.exp_name, train_freq=1000))
        task.use(termination_checker(max_env_step=int(3e5)))
        task.run(max_step=max_step)
    collector_env.close()
    evaluator_env.close()
    return policy


if __name__ == '__main__':
    main()
