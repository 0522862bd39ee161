exp_config = {
    "exp_name": "smoke_lunarlander_discrete_sac_seed0_260912_073535",
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
            "twin_critic": True,
            "action_space": "reparameterization",
            "obs_shape": 8,
            "action_shape": 4,
            "actor_head_hidden_size": 64,
            "critic_head_hidden_size": 64
        },
        "type": "discrete_sac_command",
        "multi_agent": False,
        "priority": False,
        "priority_IS_weight": False,
        "random_collect_size": 16,
        "transition_with_policy_data": True,
        "learn": {
            "update_per_collect": 1,
            "batch_size": 8,
            "learning_rate_q": 0.0005,
            "learning_rate_policy": 0.0005,
            "learning_rate_alpha": 0.0003,
            "target_theta": 0.005,
            "discount_factor": 0.99,
            "alpha": 0.2,
            "auto_alpha": False,
            "log_space": True,
            "ignore_done": False,
            "target_entropy": None
        },
        "collect": {
            "unroll_len": 1,
            "n_sample": 16
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
            },
            "eps": {
                "type": "exp",
                "start": 0.95,
                "end": 0.1,
                "decay": 10000
            }
        },
        "cfg_type": "DiscreteSACPolicyCommandDict"
    }
}
