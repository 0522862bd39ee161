"""1-iteration smoke sweep over the dizoo config zoo (reference pattern:
entry/tests/test_serial_entry.py:7-57). Every config must construct its env
+ policy and survive one collect->train iteration on CPU with shrunk sizes.
"""
import copy
import importlib

import pytest


def _shrink(main_cfg, create_cfg):
    """Scale a tuned config down to smoke size (tiny envs/batches, no eval)."""
    m = copy.deepcopy(main_cfg)
    c = copy.deepcopy(create_cfg)
    m.exp_name = 'exp/smoke_' + m.exp_name
    m.env.collector_env_num = 2
    m.env.evaluator_env_num = 1
    m.env.n_evaluator_episode = 1
    m.env.max_step = 30  # lite envs honour this cap
    p = m.policy
    p.cuda = False
    if p.get('random_collect_size', 0) > 0:
        p.random_collect_size = 16
    if 'learn' in p:
        if p.learn.get('update_per_collect') is not None:
            p.learn.update_per_collect = 1
        if p.learn.get('epoch_per_collect') is not None:
            p.learn.epoch_per_collect = 1
        if p.learn.get('batch_size') is not None:
            p.learn.batch_size = min(p.learn.batch_size, 8)
    if 'collect' in p and p.collect.get('n_sample') is not None:
        p.collect.n_sample = max(int(p.get('unroll_len', 1)) * 2, 16)
    if 'collect' in p and p.collect.get('n_episode') is not None:
        p.collect.n_episode = 2
    if 'other' in p and p.other.get('replay_buffer') is not None \
            and p.other.replay_buffer.get('replay_buffer_size') is not None:
        p.other.replay_buffer.replay_buffer_size = 1000
    # keep subprocess manager machinery out of the smoke lane
    c.env_manager.type = 'base'
    return m, c


# (config module, pipeline entry)
SMOKE_CONFIGS = [
    ('dizoo.box2d.lunarlander.config.lunarlander_dqn_config', 'serial'),
    ('dizoo.box2d.lunarlander.config.lunarlander_c51_config', 'serial'),
    ('dizoo.box2d.lunarlander.config.lunarlander_rainbow_config', 'serial'),
    ('dizoo.box2d.lunarlander.config.lunarlander_r2d2_config', 'serial'),
    ('dizoo.box2d.lunarlander.config.lunarlander_ppo_config', 'onpolicy'),
    ('dizoo.box2d.lunarlander.config.lunarlander_a2c_config', 'onpolicy'),
    ('dizoo.box2d.lunarlander.config.lunarlander_impala_config', 'serial'),
    ('dizoo.box2d.lunarlander.config.lunarlander_discrete_sac_config', 'serial'),
    ('dizoo.box2d.lunarlander.config.lunarlander_cont_sac_config', 'serial'),
    ('dizoo.box2d.lunarlander.config.lunarlander_cont_td3_config', 'serial'),
    ('dizoo.box2d.bipedalwalker.config.bipedalwalker_sac_config', 'serial'),
    ('dizoo.box2d.bipedalwalker.config.bipedalwalker_td3_config', 'serial'),
    ('dizoo.frozen_lake.config.frozen_lake_dqn_config', 'serial'),
    ('dizoo.frozen_lake.config.frozen_lake_sql_config', 'serial'),
    ('dizoo.taxi.config.taxi_dqn_config', 'serial'),
    ('dizoo.pomdp.config.pomdp_dqn_config', 'serial'),
    ('dizoo.pomdp.config.pomdp_ppo_config', 'serial'),
    ('dizoo.gym_hybrid.config.gym_hybrid_pdqn_config', 'serial'),
    ('dizoo.gym_hybrid.config.gym_hybrid_mpdqn_config', 'serial'),
    ('dizoo.gym_hybrid.config.gym_hybrid_hppo_config', 'onpolicy'),
    ('dizoo.gym_hybrid.config.gym_hybrid_ddpg_config', 'serial'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_qmix_config', 'serial'),
    ('dizoo.smac.config.smac_3s5z_qmix_config', 'serial'),
    ('dizoo.smac.config.smac_MMM_qtran_config', 'serial'),
    ('dizoo.smac.config.smac_5m6m_wqmix_config', 'serial'),
    ('dizoo.smac.config.smac_10m11m_mappo_config', 'onpolicy'),
    ('dizoo.smac.config.smac_3s5z_masac_config', 'serial'),
    ('dizoo.smac.config.smac_MMM2_madqn_config', 'serial'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_vdn_config', 'serial'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_wqmix_config', 'serial'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_qtran_config', 'serial'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_collaq_config', 'serial'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_coma_config', 'serial'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_mappo_config', 'onpolicy'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_happo_config', 'onpolicy'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_madqn_config', 'serial'),
    ('dizoo.petting_zoo.config.ptz_simple_spread_atoc_config', 'serial'),
    ('dizoo.procgen.config.coinrun_dqn_config', 'serial'),
    ('dizoo.procgen.config.coinrun_ppo_config', 'onpolicy'),
    ('dizoo.procgen.config.coinrun_ppg_config', 'onpolicy_ppg'),
    ('dizoo.procgen.config.maze_dqn_config', 'serial'),
    ('dizoo.procgen.config.maze_ppo_config', 'onpolicy'),
    ('dizoo.procgen.config.bigfish_ppg_config', 'onpolicy_ppg'),
    ('dizoo.procgen.config.bigfish_plr_config', 'plr'),
    ('dizoo.d4rl.config.hopper_medium_td3bc_config', 'offline'),
    ('dizoo.d4rl.config.hopper_medium_cql_config', 'offline'),
    ('dizoo.d4rl.config.halfcheetah_medium_iql_config', 'offline'),
    ('dizoo.d4rl.config.walker2d_medium_edac_config', 'offline'),
    ('dizoo.d4rl.config.hopper_medium_bcq_config', 'offline'),
    ('dizoo.d4rl.config.hopper_expert_dt_config', 'offline_dt'),
    ('dizoo.minigrid.config.minigrid_onppo_config', 'onpolicy'),
    ('dizoo.minigrid.config.minigrid_r2d2_config', 'serial'),
    ('dizoo.minigrid.config.minigrid_rnd_onppo_config', 'reward_model'),
    ('dizoo.minigrid.config.minigrid_icm_offppo_config', 'reward_model'),
    ('dizoo.minigrid.config.minigrid_ngu_config', 'ngu'),
    ('dizoo.slime_volley.config.slime_volley_ppo_config', 'onpolicy'),
    ('dizoo.dmc2gym.config.dmc2gym_sac_state_config', 'serial'),
    ('dizoo.dmc2gym.config.dmc2gym_sac_pixel_config', 'serial'),
    ('dizoo.dmc2gym.config.dmc2gym_ppo_config', 'onpolicy'),
    ('dizoo.cliffwalking.config.cliffwalking_dqn_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_c51_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_ppo_config', 'onpolicy'),
    ('dizoo.classic_control.cartpole.config.cartpole_a2c_config', 'onpolicy'),
    ('dizoo.classic_control.cartpole.config.cartpole_impala_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_sac_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_r2d2_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_sql_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_ppo_offpolicy_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_qrdqn_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_iqn_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_fqf_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_rainbow_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_mdqn_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_sqn_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_pg_config', 'onpolicy'),
    ('dizoo.classic_control.cartpole.config.cartpole_ppg_offpolicy_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_acer_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_dqn_stdim_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_ppo_stdim_config', 'onpolicy'),
    ('dizoo.classic_control.cartpole.config.cartpole_dqfd_config', 'dqfd'),
    ('dizoo.classic_control.pendulum.config.pendulum_ddpg_config', 'serial'),
    ('dizoo.classic_control.pendulum.config.pendulum_td3_config', 'serial'),
    ('dizoo.classic_control.pendulum.config.pendulum_sac_config', 'serial'),
    ('dizoo.classic_control.pendulum.config.pendulum_d4pg_config', 'serial'),
    ('dizoo.classic_control.pendulum.config.pendulum_td3_vae_config', 'td3_vae'),
    ('dizoo.classic_control.pendulum.config.pendulum_ibc_config', 'offline_gen'),
    ('dizoo.classic_control.cartpole.config.cartpole_discrete_cql_config', 'offline_gen'),
    ('dizoo.classic_control.pendulum.config.pendulum_mbsac_ddppo_config', 'dream'),
    ('dizoo.classic_control.pendulum.config.pendulum_stevesac_ddppo_config', 'dream'),
    ('dizoo.mujoco.config.hopper_bdq_config', 'serial'),
    ('dizoo.mujoco.config.walker2d_td3_config', 'serial'),
    ('dizoo.mujoco.config.hopper_d4pg_config', 'serial'),
    ('dizoo.mujoco.config.ant_sac_config', 'serial'),
    ('dizoo.mujoco.config.humanoid_onppo_config', 'onpolicy'),
    ('dizoo.maze.config.maze_pc_config', 'pc'),
    ('dizoo.dmc2gym.config.cartpole_balance_dreamer_config', 'dreamer'),
    ('dizoo.tabmwp.config.tabmwp_prompt_pg_config', 'onpolicy'),
    ('dizoo.classic_control.cartpole.config.cartpole_ppo_pg_config', 'onpolicy'),
    ('dizoo.classic_control.cartpole.config.cartpole_r2d2_gtrxl_config', 'serial'),
    ('dizoo.classic_control.cartpole.config.cartpole_r2d3_config', 'r2d3'),
    ('dizoo.classic_control.pendulum.config.pendulum_sqil_sac_config', 'sqil'),
    ('dizoo.league_demo.league_demo_il_config', 'serial'),
    ('dizoo.bsuite.config.deep_sea_dqn_config', 'serial'),
    ('dizoo.bsuite.config.memory_len_r2d2_config', 'serial'),
    ('dizoo.sokoban.config.sokoban_dqn_config', 'serial'),
    ('dizoo.ising_env.config.ising_mf_qmix_config', 'serial'),
    ('dizoo.gym_anytrading.config.stocks_dqn_config', 'serial'),
    ('dizoo.beergame.config.beergame_onppo_config', 'onpolicy'),
    ('dizoo.competitive_rl.config.cpong_dqn_config', 'serial'),
    ('dizoo.multiagent_mujoco.config.mamujoco_mappo_config', 'onpolicy'),
    ('dizoo.rocket.config.rocket_hover_ppo_config', 'onpolicy'),
    ('dizoo.gym_soccer.config.gym_soccer_pdqn_config', 'serial'),
    ('dizoo.pybullet.config.hopper_bullet_sac_config', 'serial'),
    ('dizoo.gym_pybullet_drones.config.drone_hover_td3_config', 'serial'),
    ('dizoo.overcooked.config.overcooked_qmix_config', 'serial'),
    ('dizoo.atari.config.serial.pong_dqn_config', 'serial'),
    ('dizoo.atari.config.serial.qbert_rainbow_config', 'serial'),
    ('dizoo.atari.config.serial.spaceinvaders_offppo_config', 'serial'),
    ('dizoo.atari.config.serial.qbert_impala_config', 'serial'),
    ('dizoo.atari.config.serial.pong_mdqn_config', 'serial'),
    ('dizoo.mario.config.mario_dqn_config', 'serial'),
    ('dizoo.metadrive.config.metadrive_onppo_config', 'onpolicy'),
    ('dizoo.evogym.config.walker_ppo_config', 'onpolicy'),
    ('dizoo.gfootball.config.gfootball_academy_ppo_config', 'serial'),
]


def _run_one(module_name: str, pipeline: str, tmp_dir: str = None):
    mod = importlib.import_module(module_name)
    m, c = _shrink(mod.main_config, mod.create_config)
    if module_name.startswith('dizoo.d4rl'):
        import os
        from dizoo.d4rl.generate import generate_d4rl_npz
        path = os.path.join(tmp_dir or '.', m.env.env_id + '.npz')
        generate_d4rl_npz(m.env.env_id, path, n_transitions=512)
        if 'dataset' in m:
            m.dataset.data_dir_prefix = path
        if m.policy.get('collect', {}).get('data_path'):
            m.policy.collect.data_path = path
    if pipeline == 'offline_dt':
        # DT trains through the task-pipeline trainer on the trajectory dataset
        import torch
        from ding.utils.data.dataset import D4RLTrajectoryDataset
        from ding.framework import OfflineRLContext, task
        from ding.framework.middleware import offline_data_fetcher, trainer
        from ding.policy import create_policy
        from ding.policy.dt import DTPolicy
        from ding.utils import EasyDict, deep_merge_dicts
        pcfg = EasyDict(deep_merge_dicts(DTPolicy.default_config(), m.policy))
        pcfg.type = c.policy.type
        pcfg.cuda = False
        pcfg.learn.batch_size = 8
        m.policy = pcfg
        dataset = D4RLTrajectoryDataset(m)
        policy = create_policy(pcfg, enable_field=['learn'])
        with task.start(ctx=OfflineRLContext()):
            task.use(offline_data_fetcher(m, dataset))
            task.use(trainer(m, policy.learn_mode))
            task.run(max_step=2)
        return
    if pipeline == 'onpolicy':
        from ding.entry import serial_pipeline_onpolicy
        serial_pipeline_onpolicy((m, c), seed=0, max_train_iter=1)
    elif pipeline == 'onpolicy_ppg':
        from ding.entry import serial_pipeline_onpolicy_ppg
        serial_pipeline_onpolicy_ppg((m, c), seed=0, max_train_iter=1)
    elif pipeline == 'plr':
        from ding.entry import serial_pipeline_plr
        serial_pipeline_plr((m, c), seed=0, max_train_iter=1)
    elif pipeline == 'reward_model':
        from ding.entry import serial_pipeline_reward_model
        serial_pipeline_reward_model((m, c), seed=0, max_train_iter=1)
    elif pipeline == 'ngu':
        from ding.entry import serial_pipeline_ngu
        serial_pipeline_ngu((m, c), seed=0, max_train_iter=1)
    elif pipeline == 'r2d3':
        from ding.entry import serial_pipeline_r2d3
        import copy as _copy
        em, ec = _copy.deepcopy(m), _copy.deepcopy(c)
        ec.policy.type = 'r2d2'
        serial_pipeline_r2d3((m, c), (em, ec), seed=0, max_train_iter=1)
    elif pipeline == 'sqil':
        from ding.entry import serial_pipeline_sqil
        import copy as _copy
        em, ec = _copy.deepcopy(m), _copy.deepcopy(c)
        ec.policy.type = 'sac'
        em.policy.random_collect_size = 16
        serial_pipeline_sqil((m, c), (em, ec), seed=0, max_train_iter=1)
    elif pipeline == 'td3_vae':
        from ding.entry import serial_pipeline_td3_vae
        m.policy.learn.warm_up_update = 8
        m.policy.random_collect_size = 16
        serial_pipeline_td3_vae((m, c), seed=0, max_train_iter=1)
    elif pipeline == 'offline_gen':
        # synthesize a matching-shape npz for offline policies
        import os
        import numpy as np
        from ding.entry import serial_pipeline_offline
        path = os.path.join(tmp_dir or '.', 'gen.npz')
        rng = np.random.RandomState(0)
        obs_shape = m.policy.model.obs_shape
        obs_dim = obs_shape if isinstance(obs_shape, int) else int(np.prod(obs_shape))
        n = 256
        act_shape = m.policy.model.action_shape
        discrete = c.policy.type in ('discrete_cql', 'bc')
        action = rng.randint(0, act_shape, n) if discrete else \
            np.tanh(rng.randn(n, act_shape)).astype(np.float32)
        np.savez(path, obs=rng.randn(n, obs_dim).astype(np.float32), action=action,
                 reward=rng.randn(n).astype(np.float32), done=(rng.rand(n) < 0.02),
                 next_obs=rng.randn(n, obs_dim).astype(np.float32))
        m.policy.collect.data_path = path
        serial_pipeline_offline((m, c), seed=0, max_train_iter=1)
    elif pipeline == 'dream':
        from ding.entry import serial_pipeline_dream
        m.world_model.model.batch_size = 16
        m.world_model.model.ensemble_size = 2
        m.world_model.model.elite_size = 1
        if m.policy.learn.get('ensemble_size') is not None:
            m.policy.learn.ensemble_size = 2
        m.world_model.model.hidden_size = 32
        m.world_model.train_freq = 4
        m.world_model.eval_freq = 100
        m.policy.random_collect_size = 24
        serial_pipeline_dream((m, c), seed=0, max_train_iter=1)
    elif pipeline == 'dreamer':
        from ding.entry import serial_pipeline_dreamer
        m.policy.random_collect_size = 24
        m.policy.imag_horizon = 4
        m.policy.model.update(dict(dyn_stoch=8, dyn_deter=32, dyn_discrete=8, units=32,
                                   actor_layers=1, value_layers=1))
        m.policy.learn.update(dict(batch_size=4, batch_length=6))
        m.policy.collect.n_sample = 16
        m.world_model.pretrain = 1
        m.world_model.model.update(dict(dyn_stoch=8, dyn_deter=32, dyn_hidden=32, dyn_discrete=8,
                                        units=32, reward_layers=1, discount_layers=1,
                                        image_dec_layers=1, batch_size=4, batch_length=6,
                                        encoder_hidden_size_list=[32, 32]))
        serial_pipeline_dreamer((m, c), seed=0, max_train_iter=1)
    elif pipeline == 'pc':
        # BFS expert batches from the maze env (reference maze PC data path)
        import numpy as np
        import torch
        from ding.entry import serial_pipeline_pc
        from dizoo.maze.envs.maze_env import MazeEnv
        from ding.utils.misc_helpers import get_vi_sequence

        def dataset_fn():
            env = MazeEnv({'size': m.env.size})
            env.seed(0)
            obs = env.reset()
            values, _ = get_vi_sequence(env, obs)
            final_v = values[-1]
            n = m.env.size
            batch = []
            for r in range(n):
                for c2 in range(n):
                    if env.maze[r, c2] != 0 or final_v[r, c2] == -np.inf:
                        continue
                    best_a, best_val = None, final_v[r, c2]
                    for a, (dr, dc) in enumerate([(-1, 0), (1, 0), (0, -1), (0, 1)]):
                        nr, nc = r + dr, c2 + dc
                        if 0 <= nr < n and 0 <= nc < n and env.maze[nr, nc] == 0 \
                                and final_v[nr, nc] > best_val:
                            best_a, best_val = a, final_v[nr, nc]
                    if best_a is None:
                        continue
                    env._agent = (r, c2)
                    batch.append({'obs': torch.as_tensor(env._obs()), 'action': torch.tensor([best_a]),
                                  'next_obs': torch.as_tensor(env._obs()),
                                  'reward': torch.tensor([0.0]), 'done': False})
            yield batch[:32]

        m.policy.learn.dataset_fn = dataset_fn
        m.policy.learn.train_epoch = 1
        serial_pipeline_pc((m, c), seed=0, max_iter=2)
    elif pipeline == 'dqfd':
        # expert = a fresh DQN on the same env (1-iter smoke scale)
        from ding.entry import serial_pipeline_dqfd, collect_demo_data
        import copy as _copy
        em, ec = _copy.deepcopy(m), _copy.deepcopy(c)
        ec.policy.type = 'dqn'
        import os as _os
        expert_path = _os.path.join(tmp_dir or '.', 'dqfd_expert.pkl')
        collect_demo_data((em, ec), seed=0, collect_count=32, expert_data_path=expert_path)
        serial_pipeline_dqfd((m, c), expert_path, seed=0, max_train_iter=1)
    elif pipeline == 'offline':
        from ding.entry import serial_pipeline_offline
        serial_pipeline_offline((m, c), seed=0, max_train_iter=1)
    else:
        from ding.entry import serial_pipeline
        serial_pipeline((m, c), seed=0, max_train_iter=1)


@pytest.mark.parametrize('module_name,pipeline', SMOKE_CONFIGS)
def test_dizoo_config_smoke(module_name, pipeline, tmp_path):
    _run_one(module_name, pipeline, tmp_dir=str(tmp_path))


def test_slime_volley_agent_vs_agent_battle():
    """agent_vs_agent mode feeds the 1v1 battle collector (league lane)."""
    import numpy as np
    from dizoo.slime_volley.envs.slime_volley_env import SlimeVolleyEnv
    env = SlimeVolleyEnv({'agent_vs_agent': True, 'max_step': 200})
    env.seed(0)
    obs = env.reset()
    assert isinstance(obs, list) and obs[0].shape == (12, ) and obs[1].shape == (12, )
    for _ in range(200):
        ts = env.step([env.random_action(), env.random_action()])
        assert ts.reward.shape == (2, )
        assert abs(float(ts.reward[0]) + float(ts.reward[1])) < 1e-6  # zero-sum
        if ts.done:
            assert ts.info['result'] in ('wins', 'losses', 'draws')
            break
    assert ts.done


def test_dmc2gym_pixel_render_tracks_state():
    import numpy as np
    from dizoo.dmc2gym.envs.dmc2gym_env import DMC2GymEnv
    env = DMC2GymEnv({'domain_name': 'cartpole', 'task_name': 'swingup', 'from_pixels': True, 'frame_skip': 2})
    env.seed(0)
    obs = env.reset()
    assert obs.shape == (3, 84, 84) and obs.dtype == np.float32
    ts = env.step(np.array([1.0]))
    assert ts.obs.shape == (3, 84, 84)
    assert 0.0 <= float(ts.reward[0]) <= 1.0
    # swingup starts hanging: reward must be near zero, upright near one
    env2 = DMC2GymEnv({'domain_name': 'cartpole', 'task_name': 'balance'})
    env2.seed(0)
    env2.reset()
    r_bal = float(env2.step(np.array([0.0])).reward[0])
    assert r_bal > 0.5


def test_cliffwalking_env_optimal_path():
    """Greedy optimal policy scores -13 (the stop_value)."""
    import numpy as np
    from dizoo.cliffwalking.envs.cliffwalking_env import CliffWalkingEnv
    env = CliffWalkingEnv({})
    env.seed(0)
    env.reset()
    ret = 0.0
    for a in [0] + [1] * 11 + [2]:  # up, 11x right, down
        ts = env.step(np.array([a]))
        ret += float(ts.reward[0])
    assert ts.done and abs(ret - (-13.0)) < 1e-6


def test_maze_bc_with_bfs_expert():
    """Maze + value-iteration BFS expert -> one BC learn epoch (the
    procedure-cloning data path, reference dizoo/maze/config/maze_bc)."""
    import numpy as np
    import torch
    from dizoo.maze.envs.maze_env import MazeEnv
    from ding.utils.misc_helpers import get_vi_sequence
    from ding.policy import create_policy
    from ding.utils import EasyDict, deep_merge_dicts
    from ding.policy.offline import BehaviourCloningPolicy

    env = MazeEnv({'size': 9})
    env.seed(1)
    obs = env.reset()
    values, _ = get_vi_sequence(env, obs)
    final_v = values[-1]
    # expert: from every open cell, act toward the neighbour with higher value
    data = []
    for r in range(9):
        for c in range(9):
            if env.maze[r, c] != 0 or final_v[r, c] == -np.inf:
                continue
            best_a, best_val = None, final_v[r, c]
            for a, (dr, dc) in enumerate([(-1, 0), (1, 0), (0, -1), (0, 1)]):
                nr, nc = r + dr, c + dc
                if 0 <= nr < 9 and 0 <= nc < 9 and env.maze[nr, nc] == 0 and final_v[nr, nc] > best_val:
                    best_a, best_val = a, final_v[nr, nc]
            if best_a is None:
                continue
            env._agent = (r, c)
            data.append({
                'obs': torch.as_tensor(env._obs()).reshape(-1),
                'next_obs': torch.as_tensor(env._obs()).reshape(-1),
                'action': torch.tensor([best_a]),
                'reward': torch.tensor([0.0]),
                'done': False,
            })
    assert len(data) > 10
    cfg = EasyDict(deep_merge_dicts(BehaviourCloningPolicy.default_config(), EasyDict(dict(
        cuda=False, continuous=False,
        model=dict(obs_shape=8 * 9 * 9, action_shape=4, encoder_hidden_size_list=[64, 64]),
        learn=dict(batch_size=16, learning_rate=1e-3),
    ))))
    pol = create_policy(cfg, enable_field=['learn'])
    out = pol._forward_learn(data[:16])
    assert 'total_loss' in out or 'loss' in out


def test_image_classification_supervised():
    """Supervised lane: CE training + metric evaluation accuracy improves on
    the synthetic digits (reference dizoo/image_classification)."""
    import torch
    from torch.utils.data import DataLoader
    from dizoo.image_classification.policy import ImageClassificationPolicy
    from dizoo.image_classification.data.dataset import SyntheticDigits
    from ding.utils import EasyDict, deep_merge_dicts
    cfg = EasyDict(deep_merge_dicts(ImageClassificationPolicy.default_config(), EasyDict(dict(
        cuda=False,
        model=dict(obs_shape=[1, 28, 28], action_shape=10, encoder_hidden_size_list=[16, 16, 32]),
        learn=dict(batch_size=64, learning_rate=1e-3),
    ))))
    pol = ImageClassificationPolicy(cfg, enable_field=['learn', 'eval'])
    ds = SyntheticDigits(n=512, seed=3)
    loader = DataLoader(ds, batch_size=64, shuffle=True, collate_fn=lambda b: b)
    accs = []
    for epoch in range(8):
        for batch in loader:
            out = pol._forward_learn(batch)
        accs.append(out['acc'])
    assert accs[-1] > 0.5, f"classifier failed to learn: {accs}"


def test_multi_discrete_dqn_policy():
    """dizoo/common md_dqn: per-branch TD over a list of q heads."""
    import torch
    from dizoo.common.policy.md_policies import MultiDiscreteDQNPolicy
    from ding.utils import EasyDict, deep_merge_dicts
    cfg = EasyDict(deep_merge_dicts(MultiDiscreteDQNPolicy.default_config(), EasyDict(dict(
        cuda=False, nstep=1,
        model=dict(obs_shape=8, action_shape=[2, 3], encoder_hidden_size_list=[16, 16]),
        learn=dict(batch_size=8, update_per_collect=1, learning_rate=1e-3),
    ))))
    pol = MultiDiscreteDQNPolicy(cfg, enable_field=['learn'])
    data = [dict(obs=torch.randn(8), next_obs=torch.randn(8), action=torch.tensor([0, 2]),
                 reward=torch.tensor([1.0]), done=False) for _ in range(8)]
    losses = [pol._forward_learn([dict(d) for d in data])['total_loss'] for _ in range(5)]
    assert losses[-1] < losses[0], losses
