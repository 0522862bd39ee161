from .bipedalwalker_env import BipedalWalkerEnv
