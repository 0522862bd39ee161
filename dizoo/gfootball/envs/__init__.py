from .gfootball_env import GFootballAcademyEnv
