"""pdqn.py example (reference ding/example/pdqn.py): runs the gym_hybrid_pdqn_config
config through serial_pipeline."""
from ding.entry import serial_pipeline


def main(max_train_iter: int = 100, seed: int = 0):
    from dizoo.gym_hybrid.config.gym_hybrid_pdqn_config import main_config, create_config
    import copy
    m = copy.deepcopy(main_config)
    m.exp_name = 'exp/example_pdqn'
    return serial_pipeline((m, copy.deepcopy(create_config)), seed=seed,
                   max_train_iter=max_train_iter)


if __name__ == '__main__':
    main()
