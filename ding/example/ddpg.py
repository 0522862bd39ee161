"""ddpg.py middleware example (reference ding/example/ddpg.py)."""
from ding.policy import DDPGPolicy
from .common import pendulum_envs, offpolicy_main


def main(max_step: int = 1000):
    return offpolicy_main('dizoo.classic_control.pendulum.config.pendulum_ddpg_config', DDPGPolicy, envs_fn=pendulum_envs, max_step=max_step,
                          use_nstep=False, use_eps=False)


if __name__ == '__main__':
    main()
