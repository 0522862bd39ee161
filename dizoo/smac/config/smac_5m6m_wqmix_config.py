"""SMAC 5m6m wqmix (reference dizoo/smac/config/smac_5m6m_wqmix_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('5m6m', 'wqmix')
smac_5m6m_wqmix_config = main_config
smac_5m6m_wqmix_create_config = create_config
