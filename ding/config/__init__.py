from .config import read_config, compile_config, compile_config_parallel, save_config, save_config_py
