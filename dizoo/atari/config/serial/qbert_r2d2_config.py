"""qbert r2d2 (reference dizoo/atari/config/serial/qbert/qbert_r2d2_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('qbert', 'r2d2')
qbert_r2d2_config = main_config
qbert_r2d2_create_config = create_config
