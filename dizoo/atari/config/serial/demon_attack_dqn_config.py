"""demon_attack dqn (reference dizoo/atari/config/serial/demon_attack/demon_attack_dqn_config.py;
built by the shared factory — see atari_family.py)."""
from dizoo.atari.config.serial.atari_family import build_atari_config

main_config, create_config = build_atari_config('demon_attack', 'dqn')
demon_attack_dqn_config = main_config
demon_attack_dqn_create_config = create_config
