"""walker2d sac (reference dizoo/mujoco/config/walker2d_sac_config.py; built by the
shared factory — see mujoco_family.py)."""
from dizoo.mujoco.config.mujoco_family import build_mujoco_config

main_config, create_config = build_mujoco_config('walker2d', 'sac')
walker2d_sac_config = main_config
walker2d_sac_create_config = create_config
