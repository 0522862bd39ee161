"""SMAC 3s5zvs3s6z madqn (reference dizoo/smac/config/smac_3s5zvs3s6z_madqn_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('3s5zvs3s6z', 'madqn')
smac_3s5zvs3s6z_madqn_config = main_config
smac_3s5zvs3s6z_madqn_create_config = create_config
