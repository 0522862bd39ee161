"""dqn_nstep.py middleware example (reference ding/example/dqn_nstep.py)."""
from ding.policy import DQNPolicy
from .common import cartpole_envs, offpolicy_main


def main(max_step: int = 1000):
    return offpolicy_main('dizoo.classic_control.cartpole.config.cartpole_dqn_config', DQNPolicy, envs_fn=cartpole_envs, max_step=max_step,
                          use_nstep=True, use_eps=True)


if __name__ == '__main__':
    main()
