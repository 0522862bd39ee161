"""qbert c51 (reference dizoo/atari/config/serial/qbert/qbert_c51_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('qbert', 'c51')
qbert_c51_config = main_config
qbert_c51_create_config = create_config
