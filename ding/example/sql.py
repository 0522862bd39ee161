"""sql.py middleware example (reference ding/example/sql.py)."""
from ding.policy import SQLPolicy
from .common import cartpole_envs, offpolicy_main


def main(max_step: int = 1000):
    return offpolicy_main('dizoo.classic_control.cartpole.config.cartpole_sql_config', SQLPolicy, envs_fn=cartpole_envs, max_step=max_step,
                          use_nstep=False, use_eps=True)


if __name__ == '__main__':
    main()
