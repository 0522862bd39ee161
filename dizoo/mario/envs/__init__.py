from .mario_env import MarioLiteEnv
