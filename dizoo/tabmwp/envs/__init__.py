from .tabmwp_env import TabMWPLiteEnv
