"""Pong DQN, multi-GPU DDP variant (reference
dizoo/atari/config/serial/pong/pong_dqn_ddp_config.py): one rank per MI355X
over RCCL/xGMI, gradients through the bucketed reducer
(ding/parallel/grad_bucket.py). Launch:

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \\
        --master-addr 127.0.0.1 dizoo/atari/example/atari_dqn_ddp.py
"""
import copy

from ding.utils import EasyDict
from dizoo.atari.config.serial.pong_dqn_config import pong_dqn_config, pong_dqn_create_config

pong_dqn_ddp_config = EasyDict(copy.deepcopy(pong_dqn_config))
pong_dqn_ddp_config.exp_name = 'pong_dqn_ddp_seed0'
pong_dqn_ddp_config.policy.multi_gpu = True
main_config = pong_dqn_ddp_config
create_config = EasyDict(copy.deepcopy(pong_dqn_create_config))
