from .base_env import BaseEnv, BaseEnvTimestep, get_vec_env_setting, get_env_cls, create_env
from .ding_env_wrapper import DingEnvWrapper
from .env_implementation_check import (check_env_implementation, check_space_dtype, check_array_space, check_reset, check_step, check_different_memory, check_obs_deepcopy, check_all, demonstrate_correct_procedure)
from .default_wrapper import get_default_wrappers
