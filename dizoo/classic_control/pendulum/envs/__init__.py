from .pendulum_env import PendulumEnv
