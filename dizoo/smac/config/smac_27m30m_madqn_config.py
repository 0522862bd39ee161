"""SMAC 27m30m madqn (reference dizoo/smac/config/smac_27m30m_madqn_config.py; built by the
shared factory — see smac_family.py)."""
from dizoo.smac.config.smac_family import build_smac_config

main_config, create_config = build_smac_config('27m30m', 'madqn')
smac_27m30m_madqn_config = main_config
smac_27m30m_madqn_create_config = create_config
