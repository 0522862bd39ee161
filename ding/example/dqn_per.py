"""CartPole DQN + prioritized replay middleware (reference
ding/example/dqn_per.py)."""
from ding.data import DequeBuffer
from ding.data.buffer.middleware import PriorityExperienceReplay
from ding.framework import OnlineRLContext, task
from ding.framework.middleware import (
    OffPolicyLearner, StepCollector, data_pusher, eps_greedy_handler, interaction_evaluator,
    priority_calculator, termination_checker,
)
from ding.policy import DQNPolicy
from ding.utils import EasyDict, deep_merge_dicts
from .common import cartpole_envs, compile


def main(max_step: int = 1000, exp_name: str = 'exp/example_dqn_per'):
    from dizoo.classic_control.cartpole.config.cartpole_dqn_config import create_config, main_config
    mc = EasyDict(deep_merge_dicts(main_config, EasyDict(dict(policy=dict(priority=True,
                                                                          priority_IS_weight=True)))))
    cfg = compile(mc, create_config, exp_name)
    collector_env, evaluator_env = cartpole_envs(cfg)
    policy = DQNPolicy(cfg.policy)
    buffer_ = DequeBuffer(size=cfg.policy.other.replay_buffer.replay_buffer_size)
    buffer_.use(PriorityExperienceReplay(buffer_, IS_weight=True))
    with task.start(ctx=OnlineRLContext()):
        task.use(interaction_evaluator(cfg, policy.eval_mode, evaluator_env))
        task.use(eps_greedy_handler(cfg))
        task.use(StepCollector(cfg, policy.collect_mode, collector_env))
        task.use(data_pusher(cfg, buffer_))
        task.use(OffPolicyLearner(cfg, policy.learn_mode, buffer_))
        task.use(priority_calculator(policy))
        task.use(termination_checker(max_env_step=int(1e5)))
        task.run(max_step=max_step)
    collector_env.close()
    evaluator_env.close()
    return policy


if __name__ == '__main__':
    main()
