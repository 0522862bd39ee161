from .lunarlander_env import LunarLanderEnv
