exp_config = {
    "exp_name": "smoke_taxi_dqn_seed0_260912_073536",
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
        "type": "taxi",
        "collector_env_num": 2,
        "evaluator_env_num": 1,
        "n_evaluator_episode": 1,
        "env_id": "Taxi-v3",
        "max_episode_steps": 60,
        "stop_value": 20,
        "max_step": 30,
        "import_names": [
            "dizoo.taxi.envs.taxi_env"
        ]
    },
    "policy": {
        "on_policy": False,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "obs_shape": 34,
            "action_shape": 6,
            "encoder_hidden_size_list": [
                128,
                128
            ]
        },
        "type": "dqn_command",
        "priority": False,
        "priority_IS_weight": False,
        "discount_factor": 0.99,
        "nstep": 3,
        "learn": {
            "update_per_collect": 1,
            "batch_size": 8,
            "learning_rate": 0.0001,
            "target_update_freq": 500,
            "target_theta": 0.005,
            "ignore_done": False
        },
        "collect": {
            "n_sample": 16,
            "unroll_len": 1
        },
        "eval": {
            "evaluator": {
                "eval_freq": 1000,
                "stop_value": 20,
                "n_episode": 1
            }
        },
        "other": {
            "eps": {
                "type": "exp",
                "start": 1,
                "end": 0.05,
                "decay": 3000000
            },
            "replay_buffer": {
                "replay_buffer_size": 1000
            }
        },
        "cfg_type": "DQNPolicyCommandDict"
    }
}
