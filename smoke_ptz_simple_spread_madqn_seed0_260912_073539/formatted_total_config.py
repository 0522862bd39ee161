exp_config = {
    "exp_name": "smoke_ptz_simple_spread_madqn_seed0_260912_073539",
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
        "on_policy": False,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "agent_num": 3,
            "obs_shape": 18,
            "global_obs_shape": 30,
            "action_shape": 5,
            "mixer": False,
            "hidden_size_list": [
                256,
                256
            ]
        },
        "type": "madqn_command",
        "priority": False,
        "priority_IS_weight": False,
        "learn": {
            "update_per_collect": 1,
            "batch_size": 8,
            "learning_rate": 0.0005,
            "clip_value": 100,
            "target_update_theta": 0.002,
            "discount_factor": 0.99,
            "double_q": False
        },
        "collect": {
            "n_sample": 16,
            "unroll_len": 16,
            "env_num": 8
        },
        "eval": {
            "env_num": 8,
            "evaluator": {
                "eval_freq": 100,
                "stop_value": 0,
                "n_episode": 1
            }
        },
        "other": {
            "eps": {
                "type": "exp",
                "start": 1.0,
                "end": 0.05,
                "decay": 100000
            },
            "replay_buffer": {
                "replay_buffer_size": 1000
            }
        },
        "cfg_type": "MADQNPolicyCommandDict",
        "nstep": 3
    }
}
