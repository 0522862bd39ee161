exp_config = {
    "exp_name": "smoke_lunarlander_impala_seed0_260912_073534",
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
            "action_shape": 4
        },
        "type": "impala_command",
        "priority": False,
        "priority_IS_weight": False,
        "unroll_len": 32,
        "transition_with_policy_data": True,
        "action_space": "discrete",
        "learn": {
            "update_per_collect": 1,
            "batch_size": 8,
            "learning_rate": 0.0003,
            "grad_clip_type": None,
            "clip_value": None,
            "optim": "adam",
            "value_weight": 0.5,
            "entropy_weight": 0.01,
            "discount_factor": 0.99,
            "lambda_": 0.95,
            "rho_clip_ratio": 1.0,
            "c_clip_ratio": 1.0,
            "rho_pg_clip_ratio": 1.0,
            "ignore_done": False
        },
        "collect": {
            "n_sample": 64,
            "collector": {
                "type": "sample"
            }
        },
        "eval": {
            "evaluator": {
                "eval_freq": 100,
                "stop_value": 200,
                "n_episode": 1
            }
        },
        "other": {
            "replay_buffer": {
                "replay_buffer_size": 1000
            }
        },
        "cfg_type": "IMPALAPolicyCommandDict"
    }
}
