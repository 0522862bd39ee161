"""SMAC MMM2 coma (reference dizoo/smac/config/smac_MMM2_coma_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('MMM2', 'coma')
smac_MMM2_coma_config = main_config
smac_MMM2_coma_create_config = create_config
