"""montezuma impala (reference dizoo/atari/config/serial/montezuma/montezuma_impala_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('montezuma', 'impala')
montezuma_impala_config = main_config
montezuma_impala_create_config = create_config
