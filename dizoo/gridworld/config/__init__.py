from .minigrid_rnd_dqn_config import main_config as minigrid_rnd_dqn_main_config, \
    create_config as minigrid_rnd_dqn_create_config
