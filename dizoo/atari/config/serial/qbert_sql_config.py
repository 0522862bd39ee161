"""qbert sql (reference dizoo/atari/config/serial/qbert/qbert_sql_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('qbert', 'sql')
qbert_sql_config = main_config
qbert_sql_create_config = create_config
