exp_config = {
    "exp_name": "smoke_lunarlander_a2c_seed0_260912_073534",
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
        "on_policy": True,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "obs_shape": 8,
            "action_shape": 4,
            "action_space": "discrete"
        },
        "type": "a2c_command",
        "priority": False,
        "priority_IS_weight": False,
        "action_space": "discrete",
        "transition_with_policy_data": True,
        "learn": {
            "batch_size": 8,
            "learning_rate": 0.0003,
            "value_weight": 0.5,
            "entropy_weight": 0.001,
            "adv_norm": True,
            "ignore_done": False,
            "grad_norm": 0.5
        },
        "collect": {
            "unroll_len": 1,
            "discount_factor": 0.99,
            "gae_lambda": 0.95,
            "n_sample": 16
        },
        "eval": {
            "evaluator": {
                "eval_freq": 100,
                "stop_value": 200,
                "n_episode": 1
            }
        },
        "cfg_type": "A2CPolicyCommandDict"
    }
}
