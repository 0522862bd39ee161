from .metadrive_env import MetaDriveLiteEnv
