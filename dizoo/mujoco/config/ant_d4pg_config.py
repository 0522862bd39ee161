"""ant d4pg (reference dizoo/mujoco/config/ant_d4pg_config.py; built by the
shared factory — see mujoco_family.py)."""
from dizoo.mujoco.config.mujoco_family import build_mujoco_config

main_config, create_config = build_mujoco_config('ant', 'd4pg')
ant_d4pg_config = main_config
ant_d4pg_create_config = create_config
