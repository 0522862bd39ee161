"""halfcheetah d4pg (reference dizoo/mujoco/config/halfcheetah_d4pg_config.py; built by the
shared factory — see mujoco_family.py)."""
from dizoo.mujoco.config.mujoco_family import build_mujoco_config

main_config, create_config = build_mujoco_config('halfcheetah', 'd4pg')
halfcheetah_d4pg_config = main_config
halfcheetah_d4pg_create_config = create_config
