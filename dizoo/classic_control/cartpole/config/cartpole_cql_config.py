"""Reference-name alias: the implementation lives in cartpole_discrete_cql_config.py
(discrete CQL)."""
from dizoo.classic_control.cartpole.config.cartpole_discrete_cql_config import *  # noqa
from dizoo.classic_control.cartpole.config.cartpole_discrete_cql_config import main_config, create_config
