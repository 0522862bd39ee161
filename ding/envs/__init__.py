from .env.base_env import BaseEnv, BaseEnvTimestep, get_vec_env_setting, get_env_cls, create_env
from .env.ding_env_wrapper import DingEnvWrapper
from .env.env_implementation_check import (check_env_implementation, check_space_dtype, check_array_space, check_reset, check_step, check_different_memory, check_obs_deepcopy, check_all, demonstrate_correct_procedure)
from .env_wrappers.env_wrappers import (
    EnvWrapper, NoopResetWrapper, MaxAndSkipWrapper, WarpFrameWrapper, ScaledFloatFrameWrapper, ClipRewardWrapper,
    FrameStackWrapper, ObsNormWrapper, RewardNormWrapper, EpisodicLifeWrapper, FireResetWrapper, TimeLimitWrapper,
    DelayRewardWrapper, EvalEpisodeReturnWrapper, ObsTransposeWrapper, FlatObsWrapper, ActionRepeatWrapper,
    update_shape,
)
from .env_manager.base_env_manager import (
    BaseEnvManager, BaseEnvManagerV2, EnvState, create_env_manager, get_env_manager_cls,
)
from .env_manager.subprocess_env_manager import (
    SyncSubprocessEnvManager, AsyncSubprocessEnvManager, SubprocessEnvManagerV2,
)
from .common.spaces import Discrete, Box, MultiDiscrete
from .env_manager.env_supervisor import EnvSupervisor
from .env.default_wrapper import get_default_wrappers
from .env_manager import setup_ding_env_manager
from . import gym_env
from .env_manager.external_managers import GymVectorEnvManager
from .env.base_env import create_model_env
