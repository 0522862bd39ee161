exp_config = {
    "exp_name": "smoke_minigrid_empty8_onppo_seed0_260912_073543",
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
        "env_id": "MiniGrid-Empty-8x8-v0",
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
        "on_policy": True,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "obs_shape": 260,
            "action_shape": 3,
            "action_space": "discrete",
            "encoder_hidden_size_list": [
                256,
                128,
                64
            ]
        },
        "type": "ppo_command",
        "priority": False,
        "priority_IS_weight": False,
        "recompute_adv": True,
        "action_space": "discrete",
        "nstep_return": False,
        "multi_agent": False,
        "transition_with_policy_data": True,
        "learn": {
            "epoch_per_collect": 1,
            "batch_size": 8,
            "learning_rate": 0.0003,
            "value_weight": 0.5,
            "entropy_weight": 0.001,
            "clip_ratio": 0.2,
            "adv_norm": True,
            "value_norm": True,
            "ppo_param_init": True,
            "grad_clip_type": "clip_norm",
            "grad_clip_value": 0.5,
            "ignore_done": False
        },
        "collect": {
            "unroll_len": 1,
            "discount_factor": 0.99,
            "gae_lambda": 0.95,
            "n_sample": 16
        },
        "eval": {
            "evaluator": {
                "eval_freq": 200,
                "stop_value": 0.96,
                "n_episode": 1
            }
        },
        "cfg_type": "PPOPolicyCommandDict"
    }
}
