"""Multi-GPU Atari DQN (reference dizoo/atari/example/atari_dqn_ddp.py):
one process per MI355X over RCCL; rank 0 evaluates, every rank collects and
learns with bucketed all-reduce gradients.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \\
        --master-addr 127.0.0.1 dizoo/atari/example/atari_dqn_ddp.py
"""
import os

import torch


def main(max_env_step: int = int(1e7), exp_name: str = None):
    from ding.entry import serial_pipeline
    from ding.utils import DDPContext
    from dizoo.atari.config.serial.pong_dqn_ddp_config import main_config, create_config
    import copy
    main, create = copy.deepcopy(main_config), copy.deepcopy(create_config)
    if exp_name:
        main.exp_name = exp_name
    with DDPContext():
        serial_pipeline((main, create), seed=0, max_env_step=max_env_step)


if __name__ == '__main__':
    main()
