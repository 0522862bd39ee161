"""pong offppo (reference dizoo/atari/config/serial/pong/pong_offppo_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('pong', 'offppo')
pong_offppo_config = main_config
pong_offppo_create_config = create_config
