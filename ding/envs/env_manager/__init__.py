from .base_env_manager import BaseEnvManager, BaseEnvManagerV2, EnvState, create_env_manager, get_env_manager_cls
from .subprocess_env_manager import SyncSubprocessEnvManager, AsyncSubprocessEnvManager, SubprocessEnvManagerV2
from .env_supervisor import EnvSupervisor
from .external_managers import PoolEnvManager, GymVectorEnvManager
from .ding_env_manager import setup_ding_env_manager
