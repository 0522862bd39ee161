from .config import (read_config, compile_config, compile_config_parallel, save_config, save_config_py,
                     Config, read_config_directly, read_config_with_system, parallel_transform,
                     parallel_transform_slurm)
from .example import A2C, C51, DDPG, DQN, PG, PPOF, PPOOffPolicy, SAC, SQL, TD3
