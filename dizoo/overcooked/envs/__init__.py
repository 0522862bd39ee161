from .overcooked_env import OvercookedLiteEnv
