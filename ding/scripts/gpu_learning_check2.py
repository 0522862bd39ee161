"""On-device learning checks beyond DQN/PPO: QMIX (cooperative MARL) and SAC
(continuous control). Each trains briefly on the MI355X and prints the
evaluator trajectory. Usage: python ding/scripts/gpu_learning_check2.py
"""
import copy
import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), '..', '..')))


def run_qmix(max_env_step: int):
    import torch
    from ding.entry import serial_pipeline
    from dizoo.smac.config.smac_3s5z_qmix_config import main_config, create_config
    from ding.utils import EasyDict
    m, c = EasyDict(copy.deepcopy(main_config)), copy.deepcopy(create_config)
    m.exp_name = 'exp/gpu_learn_qmix'
    m.policy.cuda = torch.cuda.is_available()
    m.policy.eval.evaluator.eval_freq = 50
    m.env.stop_value = 22.0  # random coop-matrix play scores ~ agent_num*25/action_dim
    serial_pipeline((m, c), seed=0, max_env_step=max_env_step)
    print('QMIX_DONE')


def run_sac(max_env_step: int):
    import torch
    from ding.entry import serial_pipeline
    from dizoo.classic_control.pendulum.config.pendulum_sac_config import main_config, create_config
    from ding.utils import EasyDict
    m, c = EasyDict(copy.deepcopy(main_config)), copy.deepcopy(create_config)
    m.exp_name = 'exp/gpu_learn_sac'
    m.policy.cuda = torch.cuda.is_available()
    m.policy.eval.evaluator.eval_freq = 200
    m.env.stop_value = -400.0  # random pendulum play scores ~ -1200
    serial_pipeline((m, c), seed=0, max_env_step=max_env_step)
    print('SAC_DONE')


if __name__ == '__main__':
    run_qmix(int(os.environ.get('QMIX_STEPS', 60000)))
    run_sac(int(os.environ.get('SAC_STEPS', 40000)))
