exp_config = {
    "exp_name": "smoke_dmc2gym_cartpole_swingup_sac_pixel_seed0_260912_073547",
    "seed": 0,
    "env": {
        "manager": {
            "episode_num": float('inf'),
            "max_retry": 1,
            "retry_type": "reset",
            "auto_reset": True,
            "step_timeout": None,
            "reset_timeout": None,
            "retry_waiting_time": 0.1,
            "shared_memory": True,
            "copy_on_get": True,
            "type": "base"
        },
        "type": "dmc2gym",
        "env_id": "dmc2gym-cartpole-swingup",
        "domain_name": "cartpole",
        "task_name": "swingup",
        "from_pixels": True,
        "channels_first": True,
        "frame_skip": 4,
        "collector_env_num": 2,
        "evaluator_env_num": 1,
        "n_evaluator_episode": 1,
        "stop_value": 180,
        "max_step": 30,
        "import_names": [
            "dizoo.dmc2gym.envs.dmc2gym_env"
        ]
    },
    "policy": {
        "on_policy": False,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "twin_critic": True,
            "action_space": "reparameterization",
            "obs_shape": [
                3,
                84,
                84
            ],
            "action_shape": 1,
            "encoder_hidden_size_list": [
                32,
                32,
                64
            ],
            "actor_head_hidden_size": 256,
            "critic_head_hidden_size": 256
        },
        "type": "sac_command",
        "multi_agent": False,
        "priority": False,
        "priority_IS_weight": False,
        "random_collect_size": 16,
        "transition_with_policy_data": True,
        "learn": {
            "update_per_collect": 1,
            "batch_size": 8,
            "learning_rate_q": 0.001,
            "learning_rate_policy": 0.001,
            "learning_rate_alpha": 0.0003,
            "target_theta": 0.005,
            "discount_factor": 0.99,
            "alpha": 0.2,
            "auto_alpha": True,
            "log_space": True,
            "ignore_done": False,
            "target_entropy": None
        },
        "collect": {
            "unroll_len": 1,
            "n_sample": 16
        },
        "eval": {
            "evaluator": {
                "eval_freq": 1000,
                "stop_value": 180,
                "n_episode": 1
            }
        },
        "other": {
            "replay_buffer": {
                "replay_buffer_size": 1000
            }
        },
        "cfg_type": "SACPolicyCommandDict"
    }
}
