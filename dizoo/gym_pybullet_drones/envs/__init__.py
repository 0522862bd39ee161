from .drone_env import DroneHoverEnv
