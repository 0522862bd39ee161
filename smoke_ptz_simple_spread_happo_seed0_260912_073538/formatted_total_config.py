exp_config = {
    "exp_name": "smoke_ptz_simple_spread_happo_seed0_260912_073538",
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
        "type": "petting_zoo",
        "env_family": "mpe",
        "env_id": "simple_spread_v2",
        "n_agent": 3,
        "n_landmark": 3,
        "max_cycles": 25,
        "agent_obs_only": False,
        "continuous_actions": False,
        "collector_env_num": 2,
        "evaluator_env_num": 1,
        "n_evaluator_episode": 1,
        "stop_value": 0,
        "max_step": 30,
        "import_names": [
            "dizoo.petting_zoo.envs.petting_zoo_simple_spread_env"
        ]
    },
    "policy": {
        "on_policy": True,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "action_space": "discrete",
            "agent_num": 3,
            "agent_obs_shape": 18,
            "global_obs_shape": 30,
            "action_shape": 5
        },
        "type": "happo_command",
        "priority": False,
        "action_space": "discrete",
        "agent_num": 3,
        "learn": {
            "epoch_per_collect": 1,
            "batch_size": 8,
            "learning_rate": 0.0005,
            "value_weight": 0.5,
            "entropy_weight": 0.01,
            "clip_ratio": 0.2,
            "adv_norm": True,
            "ignore_done": False,
            "grad_clip_type": "clip_norm",
            "grad_clip_value": 0.5
        },
        "collect": {
            "unroll_len": 1,
            "discount_factor": 0.99,
            "gae_lambda": 0.95,
            "n_sample": 16,
            "env_num": 8
        },
        "eval": {
            "env_num": 8,
            "evaluator": {
                "eval_freq": 50,
                "stop_value": 0,
                "n_episode": 1
            }
        },
        "cfg_type": "HAPPOPolicyCommandDict"
    }
}
