from .procgen_env import ProcgenEnv
