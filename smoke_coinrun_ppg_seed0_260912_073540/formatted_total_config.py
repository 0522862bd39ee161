exp_config = {
    "exp_name": "smoke_coinrun_ppg_seed0_260912_073540",
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
        "type": "procgen",
        "env_id": "coinrun",
        "collector_env_num": 2,
        "evaluator_env_num": 1,
        "n_evaluator_episode": 1,
        "stop_value": 10,
        "max_step": 30,
        "import_names": [
            "dizoo.procgen.envs.procgen_env"
        ]
    },
    "policy": {
        "on_policy": True,
        "cuda": False,
        "multi_gpu": False,
        "bp_update_sync": True,
        "traj_len_inf": False,
        "model": {
            "obs_shape": [
                3,
                64,
                64
            ],
            "action_shape": 15,
            "action_space": "discrete",
            "encoder_hidden_size_list": [
                32,
                64,
                64,
                128
            ],
            "actor_head_hidden_size": 128,
            "critic_head_hidden_size": 128
        },
        "type": "ppg_command",
        "priority": False,
        "priority_IS_weight": False,
        "recompute_adv": False,
        "action_space": "discrete",
        "nstep_return": False,
        "multi_agent": False,
        "transition_with_policy_data": True,
        "learn": {
            "epoch_per_collect": 1,
            "batch_size": 8,
            "learning_rate": 0.0005,
            "value_weight": 0.5,
            "entropy_weight": 0.01,
            "clip_ratio": 0.2,
            "adv_norm": False,
            "value_norm": True,
            "ppo_param_init": True,
            "grad_clip_type": "clip_norm",
            "grad_clip_value": 0.5,
            "ignore_done": False,
            "update_per_collect": 1,
            "aux_freq": 8,
            "aux_train_epoch": 3,
            "beta_weight": 1.0
        },
        "collect": {
            "unroll_len": 1,
            "discount_factor": 0.999,
            "gae_lambda": 0.95,
            "n_sample": 16
        },
        "eval": {
            "evaluator": {
                "eval_freq": 5000,
                "stop_value": 10,
                "n_episode": 1
            }
        },
        "other": {
            "replay_buffer": {
                "replay_buffer_size": 10000,
                "multi_buffer": True,
                "policy": {
                    "replay_buffer_size": 4096
                },
                "value": {
                    "replay_buffer_size": 32768
                }
            }
        },
        "cfg_type": "PPGPolicyCommandDict"
    }
}
