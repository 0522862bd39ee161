"""pong c51 (reference dizoo/atari/config/serial/pong/pong_c51_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('pong', 'c51')
pong_c51_config = main_config
pong_c51_create_config = create_config
