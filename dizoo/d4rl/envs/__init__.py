from .d4rl_env import D4RLLiteEnv
