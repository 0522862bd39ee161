from .sokoban_env import SokobanEnv
