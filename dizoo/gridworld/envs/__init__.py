from .minigrid_lite_env import MiniGridLiteEnv
