from .moving_env import MovingEnv
