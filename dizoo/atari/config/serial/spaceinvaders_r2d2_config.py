"""spaceinvaders r2d2 (reference dizoo/atari/config/serial/spaceinvaders/spaceinvaders_r2d2_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('spaceinvaders', 'r2d2')
spaceinvaders_r2d2_config = main_config
spaceinvaders_r2d2_create_config = create_config
