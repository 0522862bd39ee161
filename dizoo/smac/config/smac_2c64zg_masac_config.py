"""SMAC 2c64zg masac (reference dizoo/smac/config/smac_2c64zg_masac_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('2c64zg', 'masac')
smac_2c64zg_masac_config = main_config
smac_2c64zg_masac_create_config = create_config
