from .pomdp_env import PomdpLiteEnv
