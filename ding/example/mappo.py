"""mappo.py example (reference ding/example/mappo.py): runs the ptz_simple_spread_mappo_config
config through serial_pipeline_onpolicy."""
from ding.entry import serial_pipeline_onpolicy


def main(max_train_iter: int = 100, seed: int = 0):
    from dizoo.petting_zoo.config.ptz_simple_spread_mappo_config import main_config, create_config
    import copy
    m = copy.deepcopy(main_config)
    m.exp_name = 'exp/example_mappo'
    return serial_pipeline_onpolicy((m, copy.deepcopy(create_config)), seed=seed,
                   max_train_iter=max_train_iter)


if __name__ == '__main__':
    main()
