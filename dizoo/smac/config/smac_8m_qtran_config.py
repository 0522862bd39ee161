"""SMAC 8m qtran (reference dizoo/smac/config/smac_8m_qtran_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('8m', 'qtran')
smac_8m_qtran_config = main_config
smac_8m_qtran_create_config = create_config
