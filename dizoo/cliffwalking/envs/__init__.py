from .cliffwalking_env import CliffWalkingEnv
