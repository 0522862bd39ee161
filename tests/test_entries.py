"""Entry-layer tests: demo collection round-trip, eval entry, reward-model /
SQIL / GAIL / DQFD pipelines, MBRL dyna pipeline, CLI import."""
import copy
import pickle

import pytest
import torch

from ding.utils import EasyDict, deep_merge_dicts
from tests.test_policy_breadth import cartpole_cfg, pendulum_cfg


def test_collect_demo_and_eval(tmp_path):
    from ding.entry import collect_demo_data, eval as eval_entry
    main, create = cartpole_cfg('dqn')
    path = str(tmp_path / 'demo.pkl')
    data = collect_demo_data((main, create), seed=0, collect_count=32, expert_data_path=path)
    assert len(data) == 32
    with open(path, 'rb') as f:
        loaded = pickle.load(f)
    assert len(loaded) == 32
    main2, create2 = cartpole_cfg('dqn')
    value = eval_entry((main2, create2), seed=0)
    assert isinstance(value, float)


def test_reward_model_pipeline():
    from ding.entry import serial_pipeline_reward_model
    main, create = cartpole_cfg('dqn')
    main.reward_model = EasyDict(dict(type='rnd', obs_shape=4, hidden_size_list=[16, 16], update_per_collect=1))
    serial_pipeline_reward_model((main, create), seed=0, max_train_iter=2)


def test_gail_pipeline(tmp_path):
    from ding.entry import serial_pipeline_gail, collect_demo_data
    main, create = cartpole_cfg('dqn')
    expert = collect_demo_data((copy.deepcopy(main), copy.deepcopy(create)), seed=0, collect_count=16)
    main.reward_model = EasyDict(dict(type='gail', input_size=5, hidden_size=16, update_per_collect=1))
    serial_pipeline_gail((main, create), expert, seed=0, max_train_iter=2)


def test_sqil_pipeline():
    from ding.entry import serial_pipeline_sqil
    main, create = cartpole_cfg('sql', extra_policy=dict(nstep=1))
    em, ec = cartpole_cfg('sql', extra_policy=dict(nstep=1))
    serial_pipeline_sqil((main, create), (em, ec), seed=0, max_train_iter=2)


def test_dqfd_pipeline():
    from ding.entry import serial_pipeline_dqfd, collect_demo_data
    main, create = cartpole_cfg('dqn')
    expert = collect_demo_data((copy.deepcopy(main), copy.deepcopy(create)), seed=0, collect_count=16)
    main2, create2 = cartpole_cfg('dqn')
    serial_pipeline_dqfd((main2, create2), expert, seed=0, max_train_iter=2)


def test_dyna_pipeline():
    from ding.entry import serial_pipeline_dyna
    main, create = pendulum_cfg(
        'sac', extra_policy=dict(model=dict(action_space='reparameterization', twin_critic=True))
    )
    main.world_model = EasyDict(dict(
        type='mbpo', train_freq=8, eval_freq=100, cuda=False,
        model=dict(state_size=3, action_size=1, hidden_size=16, ensemble_size=2, elite_size=1, batch_size=32),
        other=dict(real_ratio=0.5, rollout_batch_size=16,
                   imagination_buffer=dict(replay_buffer_size=1000)),
        rollout_length_scheduler=dict(rollout_start_step=0, rollout_end_step=100, rollout_length_min=1,
                                      rollout_length_max=2),
    ))
    serial_pipeline_dyna((main, create), seed=0, max_train_iter=2)


def test_cli_imports():
    from ding.entry import cli, cli_ditask
    assert callable(cli) and callable(cli_ditask)


def test_atari_lite_pong_ppo_short():
    """BASELINE config #2 plumbing on CPU: tiny n_sample, base manager."""
    import copy
    from ding.entry import serial_pipeline_onpolicy
    from dizoo.atari.config.serial.pong_ppo_config import main_config, create_config
    main = copy.deepcopy(main_config)
    create = copy.deepcopy(create_config)
    main.exp_name = 'exp/test_pong_ppo_lite'
    main.policy.cuda = False
    main.env.collector_env_num = 2
    main.env.evaluator_env_num = 2
    main.env.n_evaluator_episode = 2
    main.policy.collect.n_sample = 32
    main.policy.learn.batch_size = 16
    main.policy.learn.epoch_per_collect = 1
    create.env_manager.type = 'base'
    serial_pipeline_onpolicy((main, create), seed=0, max_train_iter=2)


def test_atari_lite_impala_short():
    import copy
    from ding.entry import serial_pipeline
    from dizoo.atari.config.serial.spaceinvaders_impala_config import main_config, create_config
    main = copy.deepcopy(main_config)
    create = copy.deepcopy(create_config)
    main.exp_name = 'exp/test_impala_lite'
    main.policy.cuda = False
    main.env.collector_env_num = 2
    main.env.evaluator_env_num = 2
    main.env.n_evaluator_episode = 2
    main.policy.unroll_len = 8
    main.policy.collect.n_sample = 4
    main.policy.learn.batch_size = 2
    create.env_manager.type = 'base'
    serial_pipeline((main, create), seed=0, max_train_iter=2)


def test_r2d3_pipeline():
    from ding.entry import serial_pipeline_r2d3
    main, create = cartpole_cfg('dqn')
    main.policy.collect.pho = 0.25
    main.policy.learn.expert_replay_buffer_size = 32
    em, ec = cartpole_cfg('dqn')
    serial_pipeline_r2d3((main, create), (em, ec), seed=0, max_train_iter=2)


def test_ngu_entry_pipeline():
    from ding.entry import serial_pipeline_ngu
    main, create = cartpole_cfg('dqn')
    main.rnd_reward_model = EasyDict(dict(
        type='rnd-ngu', obs_shape=4, hidden_size_list=[16, 16], update_per_collect=1, batch_size=8
    ))
    main.episodic_reward_model = EasyDict(dict(
        type='episodic', obs_shape=4, hidden_size_list=[16, 16], update_per_collect=1, batch_size=8
    ))
    serial_pipeline_ngu((main, create), seed=0, max_train_iter=2)


def test_trex_pipeline(tmp_path):
    import pickle as pkl
    from ding.entry import serial_pipeline_preference_based_irl
    main, create = cartpole_cfg('dqn')
    # fabricate ranked demo episodes: obs sequences + returns
    episodes = [[torch.randn(4) for _ in range(8)] for _ in range(4)]
    returns = [1.0, 2.0, 3.0, 4.0]
    with open(tmp_path / 'episodes_data.pkl', 'wb') as f:
        pkl.dump(episodes, f)
    with open(tmp_path / 'learning_returns.pkl', 'wb') as f:
        pkl.dump(returns, f)
    main.reward_model = EasyDict(dict(
        type='trex', obs_shape=4, hidden_size_list=[16, 16], update_per_collect=2,
        data_path=str(tmp_path), num_snippets=8, snippet_length=4,
    ))
    serial_pipeline_preference_based_irl((main, create), seed=0, max_train_iter=2)


def test_onpolicy_ppg_pipeline():
    from ding.entry import serial_pipeline_onpolicy_ppg
    main, create = cartpole_cfg('ppg', extra_policy=dict(
        learn=dict(epoch_per_collect=1, aux_freq=1),
        collect=dict(discount_factor=0.99, gae_lambda=0.95),
    ))
    del main.policy.other  # on-policy: no replay buffer needed but _build_workers wants one
    main.policy.other = EasyDict(dict(replay_buffer=dict(type='naive', replay_buffer_size=100)))
    serial_pipeline_onpolicy_ppg((main, create), seed=0, max_train_iter=2)


def test_bco_pipeline():
    from ding.entry import serial_pipeline_bco
    main, create = cartpole_cfg('bc')
    em, ec = cartpole_cfg('dqn')
    serial_pipeline_bco((main, create), (em, ec), seed=0, max_train_iter=2)


def test_plr_pipeline():
    from ding.entry import serial_pipeline_plr
    main, create = cartpole_cfg('ppo', extra_policy=dict(
        action_space='discrete', recompute_adv=True,
        learn=dict(epoch_per_collect=1),
        collect=dict(discount_factor=0.99, gae_lambda=0.95),
    ))
    main.level_replay = EasyDict(dict(strategy='policy_entropy', num_seeds=8))
    serial_pipeline_plr((main, create), seed=0, max_train_iter=2)


def test_ding_cli_serial_subprocess(tmp_path):
    """`ding -m serial -c cfg.py -s 0` end-to-end through the console entry."""
    import subprocess
    import sys
    import textwrap
    cfg = tmp_path / 'cli_cfg.py'
    cfg.write_text(textwrap.dedent(f"""
        from ding.utils import EasyDict
        main_config = EasyDict(dict(
            exp_name='{(tmp_path / "cli_exp").as_posix()}',
            env=dict(collector_env_num=2, evaluator_env_num=1, n_evaluator_episode=1, stop_value=195,
                     max_step=30),
            policy=dict(
                cuda=False,
                model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[16, 16]),
                nstep=1, discount_factor=0.97,
                learn=dict(update_per_collect=1, batch_size=8, learning_rate=1e-3),
                collect=dict(n_sample=16),
                eval=dict(evaluator=dict(eval_freq=int(1e6))),
                other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=10000),
                           replay_buffer=dict(replay_buffer_size=1000)),
            ),
        ))
        create_config = EasyDict(dict(
            env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
            env_manager=dict(type='base'),
            policy=dict(type='dqn'),
        ))
    """))
    out = subprocess.run(
        [sys.executable, '-m', 'ding.entry.cli', '-m', 'serial', '-c', str(cfg), '-s', '0',
         '--train-iter', '2'],
        capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
