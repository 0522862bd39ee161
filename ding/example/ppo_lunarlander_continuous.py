"""ppo_lunarlander_continuous.py example (reference ding/example/ppo_lunarlander_continuous.py): runs the lunarlander_cont_sac_config
config through serial_pipeline.
continuous LunarLander (SAC config; swap in a continuous-PPO config for strict PPO)"""
from ding.entry import serial_pipeline


def main(max_train_iter: int = 100, seed: int = 0):
    from dizoo.box2d.lunarlander.config.lunarlander_cont_sac_config import main_config, create_config
    import copy
    m = copy.deepcopy(main_config)
    m.exp_name = 'exp/example_ppo_lunarlander_continuous'
    return serial_pipeline((m, copy.deepcopy(create_config)), seed=seed,
                   max_train_iter=max_train_iter)


if __name__ == '__main__':
    main()
