exp_config = {
    "exp_name": "smoke_hopper_medium_bcq_config_seed0_260912_073542",
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
        "type": "d4rl",
        "env_id": "hopper-medium-v2",
        "collector_env_num": 2,
        "evaluator_env_num": 1,
        "n_evaluator_episode": 1,
        "stop_value": 6000,
        "use_act_scale": True,
        "max_step": 30,
        "import_names": [
            "dizoo.d4rl.envs.d4rl_env"
        ]
    },
    "policy": {
        "on_policy": False,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "obs_shape": 11,
            "action_shape": 3,
            "action_space": "regression",
            "twin_critic": True,
            "actor_head_hidden_size": 512,
            "critic_head_hidden_size": 512
        },
        "type": "bcq_command",
        "learn": {
            "batch_size": 8,
            "learning_rate_q": 0.0003,
            "learning_rate_policy": 0.0003,
            "learning_rate_vae": 0.0003,
            "lmbda": 0.75,
            "phi": 0.05,
            "target_theta": 0.005,
            "discount_factor": 0.99,
            "update_per_collect": 1
        },
        "collect": {
            "unroll_len": 1,
            "data_type": "hdf5",
            "data_path": "/tmp/pytest-of-root/pytest-94/test_dizoo_config_smoke_dizoo_38/hopper-medium-v2.npz",
            "normalize_states": True
        },
        "eval": {
            "evaluator": {
                "eval_freq": 1000,
                "stop_value": 6000,
                "n_episode": 1
            }
        },
        "other": {
            "replay_buffer": {
                "replay_buffer_size": 10000
            }
        },
        "cfg_type": "BCQPolicyCommandDict"
    }
}
