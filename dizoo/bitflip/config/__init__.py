from .bitflip_her_dqn_config import main_config as bitflip_her_dqn_main_config, \
    create_config as bitflip_her_dqn_create_config
