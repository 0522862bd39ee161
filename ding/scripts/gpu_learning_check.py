"""On-device learning check: train pong_dqn (atari-lite) on the MI355X for a
bounded number of env steps and report the evaluator's return trajectory —
evidence the GPU training loop learns, not merely steps.
Usage: python ding/scripts/gpu_learning_check.py [max_env_step]
"""
import copy
import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), '..', '..')))


def main():
    import torch
    algo = sys.argv[2] if len(sys.argv) > 2 else 'dqn'
    max_env_step = int(sys.argv[1]) if len(sys.argv) > 1 else 60000
    if algo == 'ppo':
        from ding.entry import serial_pipeline_onpolicy as pipeline
        from dizoo.atari.config.serial.pong_ppo_config import create_config, main_config
    else:
        from ding.entry import serial_pipeline as pipeline
        from dizoo.atari.config.serial.pong_dqn_config import create_config, main_config
    m, c = copy.deepcopy(main_config), copy.deepcopy(create_config)
    m = type(m)(m) if isinstance(m, dict) else m
    from ding.utils import EasyDict
    m = EasyDict(m)
    m.exp_name = f'exp/gpu_learning_check_{algo}'
    m.policy.cuda = torch.cuda.is_available()
    m.env.collector_env_num = 8
    m.env.evaluator_env_num = 4
    m.env.n_evaluator_episode = 4
    if algo == 'ppo':
        m.policy.collect.n_sample = 1024
        m.policy.learn.batch_size = 256
        m.policy.eval.evaluator.eval_freq = 2
    else:
        m.policy.other.eps.decay = 20000
        m.policy.eval.evaluator.eval_freq = 500
    # random play scores ~0.125/step; a learned policy approaches ~1/step
    m.env.stop_value = 250.0
    policy = pipeline((m, c), seed=0, max_env_step=max_env_step)
    print('LEARNING_CHECK_DONE')


if __name__ == '__main__':
    main()
