"""SMAC 8m qmix (reference dizoo/smac/config/smac_8m_qmix_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('8m', 'qmix')
smac_8m_qmix_config = main_config
smac_8m_qmix_create_config = create_config
