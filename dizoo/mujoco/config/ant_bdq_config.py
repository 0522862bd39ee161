"""ant bdq (reference dizoo/mujoco/config/ant_bdq_config.py; built by the
shared factory — see mujoco_family.py)."""
from dizoo.mujoco.config.mujoco_family import build_mujoco_config

main_config, create_config = build_mujoco_config('ant', 'bdq')
ant_bdq_config = main_config
ant_bdq_create_config = create_config
