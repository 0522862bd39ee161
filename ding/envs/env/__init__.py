from .base_env import BaseEnv, BaseEnvTimestep, get_vec_env_setting, get_env_cls, create_env
from .ding_env_wrapper import DingEnvWrapper
from .env_implementation_check import check_env_implementation
from .default_wrapper import get_default_wrappers
