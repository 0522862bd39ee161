"""asterix impala (reference dizoo/atari/config/serial/asterix/asterix_impala_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('asterix', 'impala')
asterix_impala_config = main_config
asterix_impala_create_config = create_config
