"""SMAC 5m6m mappo (reference dizoo/smac/config/smac_5m6m_mappo_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('5m6m', 'mappo')
smac_5m6m_mappo_config = main_config
smac_5m6m_mappo_create_config = create_config
