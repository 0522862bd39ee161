"""bcq.py example (reference ding/example/bcq.py): runs the hopper_medium_bcq_config
config through serial_pipeline_offline.
offline: generate the dataset first (dizoo/d4rl/generate.py)"""
from ding.entry import serial_pipeline_offline


def main(max_train_iter: int = 100, seed: int = 0):
    from dizoo.d4rl.config.hopper_medium_bcq_config import main_config, create_config
    import copy
    m = copy.deepcopy(main_config)
    m.exp_name = 'exp/example_bcq'
    return serial_pipeline_offline((m, copy.deepcopy(create_config)), seed=seed,
                   max_train_iter=max_train_iter)


if __name__ == '__main__':
    main()
