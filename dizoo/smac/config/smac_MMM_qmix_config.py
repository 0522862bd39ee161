"""SMAC MMM qmix (reference dizoo/smac/config/smac_MMM_qmix_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('MMM', 'qmix')
smac_MMM_qmix_config = main_config
smac_MMM_qmix_create_config = create_config
