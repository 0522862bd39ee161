from .mamujoco_env import MAMujocoEnv
