"""spaceinvaders qrdqn (reference dizoo/atari/config/serial/spaceinvaders/spaceinvaders_qrdqn_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('spaceinvaders', 'qrdqn')
spaceinvaders_qrdqn_config = main_config
spaceinvaders_qrdqn_create_config = create_config
