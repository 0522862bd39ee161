"""Reference-name alias: the implementation lives in cartpole_ppo_pg_config.py
(pure policy-gradient PPO)."""
from dizoo.classic_control.cartpole.config.cartpole_ppo_pg_config import *  # noqa
from dizoo.classic_control.cartpole.config.cartpole_ppo_pg_config import main_config, create_config
