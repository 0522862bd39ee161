"""montezuma dqn (reference dizoo/atari/config/serial/montezuma/montezuma_dqn_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('montezuma', 'dqn')
montezuma_dqn_config = main_config
montezuma_dqn_create_config = create_config
