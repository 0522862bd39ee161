from .memory_len_env import MemoryLenEnv
