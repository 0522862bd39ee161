from .coop_matrix_env import CoopMatrixEnv
