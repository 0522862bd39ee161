from .dmc2gym_env import DMC2GymEnv
