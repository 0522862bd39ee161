"""phoenix onppo (reference dizoo/atari/config/serial/phoenix/phoenix_onppo_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('phoenix', 'onppo')
phoenix_onppo_config = main_config
phoenix_onppo_create_config = create_config
