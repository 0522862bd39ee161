"""Multi-GPU Atari PPO (reference dizoo/atari/example/atari_ppo_ddp.py):
weak-scaling on-policy PPO, one rank per MI355X over RCCL/xGMI.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \\
        --master-addr 127.0.0.1 dizoo/atari/example/atari_ppo_ddp.py
"""


def main(max_env_step: int = int(1e7), exp_name: str = None):
    from ding.entry import serial_pipeline_onpolicy
    from ding.utils import DDPContext
    from dizoo.atari.config.serial.pong_ppo_config import main_config, create_config
    import copy
    main, create = copy.deepcopy(main_config), copy.deepcopy(create_config)
    main.policy.multi_gpu = True
    if exp_name:
        main.exp_name = exp_name
    with DDPContext():
        serial_pipeline_onpolicy((main, create), seed=0, max_env_step=max_env_step)


if __name__ == '__main__':
    main()
