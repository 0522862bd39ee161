from .bitflip_env import BitFlipEnv
