from .ising_model_env import IsingModelEnv
