from .memory_len_r2d2_config import main_config as memory_len_r2d2_main_config, \
    create_config as memory_len_r2d2_create_config
