"""walker2d td3 (reference dizoo/mujoco/config/walker2d_td3_config.py; built by the
shared factory — see mujoco_family.py)."""
from dizoo.mujoco.config.mujoco_family import build_mujoco_config

main_config, create_config = build_mujoco_config('walker2d', 'td3')
walker2d_td3_config = main_config
walker2d_td3_create_config = create_config
