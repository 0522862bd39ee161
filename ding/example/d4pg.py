"""d4pg.py middleware example (reference ding/example/d4pg.py)."""
from ding.policy import D4PGPolicy
from .common import pendulum_envs, offpolicy_main


def main(max_step: int = 1000):
    return offpolicy_main('dizoo.classic_control.pendulum.config.pendulum_d4pg_config', D4PGPolicy, envs_fn=pendulum_envs, max_step=max_step,
                          use_nstep=True, use_eps=False)


if __name__ == '__main__':
    main()
