"""humanoid bdq (reference dizoo/mujoco/config/humanoid_bdq_config.py; built by the
shared factory — see mujoco_family.py)."""
from dizoo.mujoco.config.mujoco_family import build_mujoco_config

main_config, create_config = build_mujoco_config('humanoid', 'bdq')
humanoid_bdq_config = main_config
humanoid_bdq_create_config = create_config
