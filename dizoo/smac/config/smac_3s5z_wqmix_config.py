"""SMAC 3s5z wqmix (reference dizoo/smac/config/smac_3s5z_wqmix_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('3s5z', 'wqmix')
smac_3s5z_wqmix_config = main_config
smac_3s5z_wqmix_create_config = create_config
