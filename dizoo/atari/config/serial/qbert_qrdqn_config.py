"""qbert qrdqn (reference dizoo/atari/config/serial/qbert/qbert_qrdqn_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('qbert', 'qrdqn')
qbert_qrdqn_config = main_config
qbert_qrdqn_create_config = create_config
