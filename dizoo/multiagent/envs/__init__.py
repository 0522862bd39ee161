from .coop_matrix_env import CoopMatrixEnv
from .particle_env import ParticleSpreadEnv
