from .rocket_env import RocketEnv
