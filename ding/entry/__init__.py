from .serial_entry import serial_pipeline, serial_pipeline_onpolicy, random_collect
from .serial_entry_offline import serial_pipeline_offline
from .serial_entry_variants import (
    serial_pipeline_reward_model, serial_pipeline_sqil, serial_pipeline_gail, serial_pipeline_dqfd,
    serial_pipeline_bc, serial_pipeline_dyna, serial_pipeline_dream,
)
from .application_entry import eval, collect_demo_data, collect_episodic_demo_data, episode_to_transitions, \
    episode_to_transitions_filter, collect_episodic_demo_data_for_trex
from .parallel_entry import parallel_pipeline
from .cli import cli
from .cli_ditask import cli_ditask
from .serial_entry_variants2 import (
    serial_pipeline_ngu, serial_pipeline_r2d3, serial_pipeline_preference_based_irl,
    serial_pipeline_preference_based_irl_onpolicy, serial_pipeline_trex, serial_pipeline_trex_onpolicy,
    serial_pipeline_guided_cost, serial_pipeline_td3_vae, serial_pipeline_onpolicy_ppg,
    serial_pipeline_bco, serial_pipeline_pc, trex_collecting_data, serial_pipeline_dreamer, serial_pipeline_plr, generate_seeds,
)
from .dist_entry import (
    dist_prepare_config, dist_launch_coordinator, dist_launch_learner, dist_launch_collector, dist_add_replicas,
    dist_delete_replicas, dist_restart_replicas,
)

# reference-spelling aliases: the single reward-model pipeline handles both
# on- and off-policy branches internally (serial_entry_variants.py)
serial_pipeline_reward_model_offpolicy = serial_pipeline_reward_model
serial_pipeline_reward_model_onpolicy = serial_pipeline_reward_model
