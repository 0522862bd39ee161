from .atari_lite_env import AtariLiteEnv
