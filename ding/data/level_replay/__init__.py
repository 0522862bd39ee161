from .level_sampler import LevelSampler
