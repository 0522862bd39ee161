from .mujoco_lite_env import MujocoLiteEnv
