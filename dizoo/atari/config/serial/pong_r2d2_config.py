"""pong r2d2 (reference dizoo/atari/config/serial/pong/pong_r2d2_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('pong', 'r2d2')
pong_r2d2_config = main_config
pong_r2d2_create_config = create_config
