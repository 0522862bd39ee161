from .beergame_env import BeerGameEnv
