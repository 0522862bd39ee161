exp_config = {
    "exp_name": "smoke_gym_hybrid_pdqn_seed0_260912_073536",
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
        "type": "gym_hybrid",
        "collector_env_num": 2,
        "evaluator_env_num": 1,
        "n_evaluator_episode": 1,
        "env_id": "Moving-v0",
        "act_scale": True,
        "stop_value": 1.8,
        "max_step": 30,
        "import_names": [
            "dizoo.gym_hybrid.envs.moving_env"
        ]
    },
    "policy": {
        "on_policy": False,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "obs_shape": 10,
            "action_shape": {
                "action_type_shape": 3,
                "action_args_shape": 2
            }
        },
        "type": "pdqn_command",
        "priority": False,
        "priority_IS_weight": False,
        "discount_factor": 0.99,
        "nstep": 1,
        "learn": {
            "update_per_collect": 1,
            "batch_size": 8,
            "learning_rate": 0.001,
            "target_update_freq": 100,
            "target_theta": 0.001,
            "ignore_done": False,
            "learning_rate_dis": 0.0003,
            "learning_rate_cont": 0.0003,
            "update_circle": 10
        },
        "collect": {
            "n_sample": 16,
            "unroll_len": 1,
            "noise_sigma": 0.1
        },
        "eval": {
            "evaluator": {
                "eval_freq": 100,
                "stop_value": 1.8,
                "n_episode": 1
            }
        },
        "other": {
            "eps": {
                "type": "exp",
                "start": 1,
                "end": 0.1,
                "decay": 100000
            },
            "replay_buffer": {
                "replay_buffer_size": 1000
            }
        },
        "cfg_type": "PDQNPolicyCommandDict"
    }
}
