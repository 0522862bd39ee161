"""dqn_frozen_lake.py example (reference ding/example/dqn_frozen_lake.py): runs the frozen_lake_dqn_config
config through serial_pipeline."""
from ding.entry import serial_pipeline


def main(max_train_iter: int = 100, seed: int = 0):
    from dizoo.frozen_lake.config.frozen_lake_dqn_config import main_config, create_config
    import copy
    m = copy.deepcopy(main_config)
    m.exp_name = 'exp/example_dqn_frozen_lake'
    return serial_pipeline((m, copy.deepcopy(create_config)), seed=seed,
                   max_train_iter=max_train_iter)


if __name__ == '__main__':
    main()
