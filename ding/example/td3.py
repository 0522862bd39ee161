"""td3.py middleware example (reference ding/example/td3.py)."""
from ding.policy import TD3Policy
from .common import pendulum_envs, offpolicy_main


def main(max_step: int = 1000):
    return offpolicy_main('dizoo.classic_control.pendulum.config.pendulum_td3_config', TD3Policy, envs_fn=pendulum_envs, max_step=max_step,
                          use_nstep=False, use_eps=False)


if __name__ == '__main__':
    main()
