"""ant td3 (reference dizoo/mujoco/config/ant_td3_config.py; built by the
shared factory — see mujoco_family.py)."""
from dizoo.mujoco.config.mujoco_family import build_mujoco_config

main_config, create_config = build_mujoco_config('ant', 'td3')
ant_td3_config = main_config
ant_td3_create_config = create_config
