exp_config = {
    "exp_name": "smoke_bipedalwalker_td3_seed0_260912_073536",
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
        "type": "bipedalwalker",
        "collector_env_num": 2,
        "evaluator_env_num": 1,
        "n_evaluator_episode": 1,
        "stop_value": 300,
        "act_scale": True,
        "max_step": 30,
        "import_names": [
            "dizoo.box2d.bipedalwalker.envs.bipedalwalker_env"
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
            "action_space": "regression",
            "obs_shape": 24,
            "action_shape": 4
        },
        "type": "td3_command",
        "priority": False,
        "priority_IS_weight": False,
        "random_collect_size": 16,
        "transition_with_policy_data": False,
        "action_space": "continuous",
        "reward_batch_norm": False,
        "multi_agent": False,
        "learn": {
            "update_per_collect": 1,
            "batch_size": 8,
            "learning_rate_actor": 0.0003,
            "learning_rate_critic": 0.0003,
            "ignore_done": False,
            "target_theta": 0.005,
            "discount_factor": 0.99,
            "actor_update_freq": 2,
            "noise": True,
            "noise_sigma": 0.2,
            "noise_range": {
                "min": -0.5,
                "max": 0.5
            }
        },
        "collect": {
            "unroll_len": 1,
            "noise_sigma": 0.1,
            "n_sample": 16
        },
        "eval": {
            "evaluator": {
                "eval_freq": 100,
                "stop_value": 300,
                "n_episode": 1
            }
        },
        "other": {
            "replay_buffer": {
                "replay_buffer_size": 1000
            }
        },
        "cfg_type": "TD3PolicyCommandDict"
    }
}
