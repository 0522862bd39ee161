"""ant ddpg (reference dizoo/mujoco/config/ant_ddpg_config.py; built by the
shared factory — see mujoco_family.py)."""
from dizoo.mujoco.config.mujoco_family import build_mujoco_config

main_config, create_config = build_mujoco_config('ant', 'ddpg')
ant_ddpg_config = main_config
ant_ddpg_create_config = create_config
