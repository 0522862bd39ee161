from .soccer_env import SoccerEnv
