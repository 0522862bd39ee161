from .minigrid_env import MiniGridEnv
