from .maze_env import MazeEnv
