from .cpong_env import CompetitivePongEnv
