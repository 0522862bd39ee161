from .cartpole_env import CartPoleEnv
