"""hopper onppo (reference dizoo/mujoco/config/hopper_onppo_config.py; built by the
shared factory — see mujoco_family.py)."""
from dizoo.mujoco.config.mujoco_family import build_mujoco_config

main_config, create_config = build_mujoco_config('hopper', 'onppo')
hopper_onppo_config = main_config
hopper_onppo_create_config = create_config
