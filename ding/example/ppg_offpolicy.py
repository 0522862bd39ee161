"""ppg_offpolicy.py middleware example (reference ding/example/ppg_offpolicy.py)."""
from ding.policy import PPGPolicy
from .common import cartpole_envs, offpolicy_main


def main(max_step: int = 1000):
    return offpolicy_main('dizoo.classic_control.cartpole.config.cartpole_ppg_offpolicy_config', PPGPolicy, envs_fn=cartpole_envs, max_step=max_step,
                          use_nstep=False, use_eps=False)


if __name__ == '__main__':
    main()
