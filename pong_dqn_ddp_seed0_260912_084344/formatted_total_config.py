exp_config = {
    "exp_name": "pong_dqn_ddp_seed0_260912_084344",
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
            "type": "subprocess"
        },
        "type": "atari_lite",
        "collector_env_num": 8,
        "evaluator_env_num": 8,
        "n_evaluator_episode": 8,
        "stop_value": 18,
        "env_id": "PongNoFrameskip-v4",
        "frame_stack": 4,
        "import_names": [
            "dizoo.atari.envs.atari_lite_env"
        ]
    },
    "policy": {
        "on_policy": False,
        "cuda": True,
        "multi_gpu": True,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "obs_shape": [
                4,
                84,
                84
            ],
            "action_shape": 6,
            "encoder_hidden_size_list": [
                128,
                128,
                512
            ]
        },
        "type": "dqn_command",
        "priority": False,
        "priority_IS_weight": False,
        "discount_factor": 0.99,
        "nstep": 3,
        "learn": {
            "update_per_collect": 10,
            "batch_size": 32,
            "learning_rate": 0.0001,
            "target_update_freq": 500,
            "target_theta": 0.005,
            "ignore_done": False
        },
        "collect": {
            "n_sample": 96,
            "unroll_len": 1
        },
        "eval": {
            "evaluator": {
                "eval_freq": 4000,
                "stop_value": 18,
                "n_episode": 8
            }
        },
        "other": {
            "eps": {
                "type": "exp",
                "start": 1.0,
                "end": 0.05,
                "decay": 250000
            },
            "replay_buffer": {
                "replay_buffer_size": 400000
            }
        },
        "cfg_type": "DQNPolicyCommandDict"
    }
}
