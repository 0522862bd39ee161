from .frozen_lake_env import FrozenLakeEnv
