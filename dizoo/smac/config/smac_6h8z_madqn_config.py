"""SMAC 6h8z madqn (reference dizoo/smac/config/smac_6h8z_madqn_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('6h8z', 'madqn')
smac_6h8z_madqn_config = main_config
smac_6h8z_madqn_create_config = create_config
