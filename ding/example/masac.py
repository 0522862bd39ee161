"""masac.py example (reference ding/example/masac.py): runs the smac_3s5z_masac_config
config through serial_pipeline."""
from ding.entry import serial_pipeline


def main(max_train_iter: int = 100, seed: int = 0):
    from dizoo.smac.config.smac_3s5z_masac_config import main_config, create_config
    import copy
    m = copy.deepcopy(main_config)
    m.exp_name = 'exp/example_masac'
    return serial_pipeline((m, copy.deepcopy(create_config)), seed=seed,
                   max_train_iter=max_train_iter)


if __name__ == '__main__':
    main()
