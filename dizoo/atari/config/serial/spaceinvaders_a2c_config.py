"""spaceinvaders a2c (reference dizoo/atari/config/serial/spaceinvaders/spaceinvaders_a2c_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('spaceinvaders', 'a2c')
spaceinvaders_a2c_config = main_config
spaceinvaders_a2c_create_config = create_config
