exp_config = {
    "exp_name": "smoke_lunarlander_dqn_seed0_260912_073534",
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
        "type": "lunarlander",
        "collector_env_num": 2,
        "evaluator_env_num": 1,
        "env_id": "LunarLander-v2",
        "n_evaluator_episode": 1,
        "stop_value": 200,
        "max_step": 30,
        "import_names": [
            "dizoo.box2d.lunarlander.envs.lunarlander_env"
        ]
    },
    "policy": {
        "on_policy": False,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "obs_shape": 8,
            "action_shape": 4,
            "encoder_hidden_size_list": [
                512,
                64
            ],
            "dueling": True
        },
        "type": "dqn_command",
        "priority": False,
        "priority_IS_weight": False,
        "discount_factor": 0.99,
        "nstep": 3,
        "learn": {
            "update_per_collect": 1,
            "batch_size": 8,
            "learning_rate": 0.001,
            "target_update_freq": 100,
            "target_theta": 0.005,
            "ignore_done": False
        },
        "collect": {
            "n_sample": 16,
            "unroll_len": 1
        },
        "eval": {
            "evaluator": {
                "eval_freq": 100,
                "stop_value": 200,
                "n_episode": 1
            }
        },
        "other": {
            "eps": {
                "type": "exp",
                "start": 0.95,
                "end": 0.1,
                "decay": 50000
            },
            "replay_buffer": {
                "replay_buffer_size": 1000
            }
        },
        "cfg_type": "DQNPolicyCommandDict"
    }
}
