from .game_env import GameEnv
