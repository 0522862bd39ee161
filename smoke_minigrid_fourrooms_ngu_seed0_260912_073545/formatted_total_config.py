exp_config = {
    "exp_name": "smoke_minigrid_fourrooms_ngu_seed0_260912_073545",
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
        "type": "minigrid",
        "env_id": "MiniGrid-FourRooms-v0",
        "collector_env_num": 2,
        "evaluator_env_num": 1,
        "n_evaluator_episode": 1,
        "stop_value": 0.96,
        "max_step": 30,
        "import_names": [
            "dizoo.minigrid.envs.minigrid_env"
        ]
    },
    "policy": {
        "on_policy": False,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "obs_shape": 680,
            "action_shape": 3,
            "encoder_hidden_size_list": [
                128,
                128,
                64
            ],
            "lstm_type": "normal"
        },
        "type": "r2d2_command",
        "priority": True,
        "priority_IS_weight": True,
        "discount_factor": 0.997,
        "nstep": 5,
        "burnin_step": 2,
        "learn_unroll_len": 38,
        "learn": {
            "update_per_collect": 1,
            "batch_size": 8,
            "learning_rate": 0.0005,
            "target_update_theta": 0.001,
            "value_rescale": True,
            "ignore_done": False
        },
        "collect": {
            "n_sample": 80,
            "env_num": 8,
            "traj_len_inf": True,
            "unroll_len": 40
        },
        "eval": {
            "env_num": 5,
            "evaluator": {
                "eval_freq": 200,
                "stop_value": 0.96,
                "n_episode": 1
            }
        },
        "other": {
            "eps": {
                "type": "exp",
                "start": 0.95,
                "end": 0.05,
                "decay": 100000
            },
            "replay_buffer": {
                "replay_buffer_size": 1000
            }
        },
        "cfg_type": "R2D2PolicyCommandDict",
        "unroll_len": 40
    },
    "rnd_reward_model": {
        "type": "rnd-ngu",
        "intrinsic_reward_type": "add",
        "obs_shape": 680,
        "hidden_size_list": [
            128,
            64
        ],
        "learning_rate": 0.0005,
        "update_per_collect": 10,
        "batch_size": 320
    },
    "episodic_reward_model": {
        "type": "episodic",
        "intrinsic_reward_type": "add",
        "obs_shape": 680,
        "action_shape": 3,
        "hidden_size_list": [
            128,
            64
        ],
        "learning_rate": 0.0005,
        "update_per_collect": 10,
        "batch_size": 320
    }
}
