exp_config = {
    "exp_name": "smoke_frozen_lake_sql_seed0_260912_073536",
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
        "type": "frozen_lake",
        "collector_env_num": 2,
        "evaluator_env_num": 1,
        "n_evaluator_episode": 1,
        "env_id": "FrozenLake-v1",
        "desc": None,
        "map_name": "4x4",
        "is_slippery": False,
        "stop_value": 0.95,
        "max_step": 30,
        "import_names": [
            "dizoo.frozen_lake.envs.frozen_lake_env"
        ]
    },
    "policy": {
        "on_policy": False,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "obs_shape": 16,
            "action_shape": 4,
            "encoder_hidden_size_list": [
                128,
                128,
                64
            ]
        },
        "type": "sql_command",
        "priority": False,
        "priority_IS_weight": False,
        "discount_factor": 0.97,
        "nstep": 1,
        "learn": {
            "update_per_collect": 1,
            "batch_size": 8,
            "learning_rate": 0.001,
            "target_update_freq": 100,
            "target_theta": 0.005,
            "ignore_done": False,
            "alpha": 0.1
        },
        "collect": {
            "n_sample": 16,
            "unroll_len": 1
        },
        "eval": {
            "evaluator": {
                "eval_freq": 40,
                "stop_value": 0.95,
                "n_episode": 1
            }
        },
        "other": {
            "eps": {
                "type": "exp",
                "start": 0.95,
                "end": 0.1,
                "decay": 10000
            },
            "replay_buffer": {
                "replay_buffer_size": 1000
            }
        },
        "cfg_type": "SQLPolicyCommandDict"
    }
}
