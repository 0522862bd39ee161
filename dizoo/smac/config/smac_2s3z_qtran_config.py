"""SMAC 2s3z qtran (reference dizoo/smac/config/smac_2s3z_qtran_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('2s3z', 'qtran')
smac_2s3z_qtran_config = main_config
smac_2s3z_qtran_create_config = create_config
