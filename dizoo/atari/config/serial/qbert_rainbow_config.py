"""qbert rainbow (reference dizoo/atari/config/serial/qbert/qbert_rainbow_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('qbert', 'rainbow')
qbert_rainbow_config = main_config
qbert_rainbow_create_config = create_config
