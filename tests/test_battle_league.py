"""Battle (1v1) collector + league job round-trip tests."""
import numpy as np
import pytest
import torch

from ding.utils import EasyDict


def _two_dqn_policies():
    from ding.policy import create_policy
    cfgs = []
    pols = []
    for _ in range(2):
        cfg = EasyDict(dict(
            type='dqn', cuda=False, on_policy=False, priority=False, priority_IS_weight=False,
            model=dict(obs_shape=4, action_shape=2, encoder_hidden_size_list=[16, 16]),
            nstep=1, discount_factor=0.99,
            learn=dict(update_per_collect=1, batch_size=8, learning_rate=1e-3,
                       target_update_freq=10, ignore_done=False),
            collect=dict(n_sample=8, unroll_len=1),
            eval=dict(),
            other=dict(eps=dict(type='exp', start=0.95, end=0.1, decay=1000),
                       replay_buffer=dict(replay_buffer_size=100)),
        ))
        p = create_policy(cfg, enable_field=['collect'])
        pols.append(p.collect_mode)
    return pols


def _battle_env_manager(env_num=2, repeat=4):
    from ding.envs import create_env_manager
    from dizoo.league_demo.game_env import GameEnv
    return create_env_manager(
        EasyDict({'type': 'base'}),
        [lambda: GameEnv({'repeat_count': repeat, 'game_type': 'zero_sum'}) for _ in range(env_num)],
    )


class _FlattenBattlePolicy:
    """Adapts a DQN collect_mode policy to the battle env's 2x2 obs."""

    def __init__(self, inner):
        self._inner = inner

    def forward(self, obs: dict, **kwargs):
        flat = {i: torch.as_tensor(o, dtype=torch.float32).flatten() for i, o in obs.items()}
        return self._inner.forward(flat, **kwargs)

    def __getattr__(self, name):
        return getattr(self._inner, name)


def test_battle_sample_collector():
    from ding.worker import BattleSampleSerialCollector
    pols = [_FlattenBattlePolicy(p) for p in _two_dqn_policies()]
    env = _battle_env_manager()
    collector = BattleSampleSerialCollector(
        EasyDict({'type': 'sample_1v1'}), env=env, policy=pols, exp_name='exp/test_battle'
    )
    data, info = collector.collect(n_sample=8, policy_kwargs={'eps': 0.5})
    assert len(data) == 2 and len(info) == 2
    for p in range(2):
        assert len(data[p]) == 8
        sample = data[p][0]
        assert 'obs' in sample and 'action' in sample and 'reward' in sample
    # rewards are per-player: p0 and p1 see different rewards for same step
    assert len(info[0]) > 0 and 'result' in info[0][0]
    collector.close()


def test_battle_episode_collector():
    from ding.worker import BattleEpisodeSerialCollector
    pols = [_FlattenBattlePolicy(p) for p in _two_dqn_policies()]
    env = _battle_env_manager()
    collector = BattleEpisodeSerialCollector(
        EasyDict({'type': 'episode_1v1'}), env=env, policy=pols, exp_name='exp/test_battle_ep'
    )
    data, info = collector.collect(n_episode=3, policy_kwargs={'eps': 0.5})
    assert len(data[0]) >= 3
    assert all(len(ep) == 4 for ep in data[0])  # repeat_count=4 steps per episode
    collector.close()


def test_one_vs_one_league_with_battle_results():
    from ding.league import create_league
    cfg = EasyDict(dict(
        league_type='one_vs_one',
        player_category=['default'],
        path_policy='exp/test_league_policy',
        active_players=dict(naive_sp_player=1),
        naive_sp_player=dict(
            one_phase_step=4,
            branch_probs=dict(pfsp=0.5, sp=0.5),
            strong_win_rate=0.7,
        ),
        use_pretrain=False,
        use_pretrain_init_historical=False,
        payoff=dict(type='battle', decay=0.99, min_win_rate_games=4),
        metric=dict(mu=0, sigma=25 / 3, beta=25 / 3 / 2, tau=0.0, draw_probability=0.02),
    ))
    league = create_league(cfg)
    pid = league.active_players_ids[0]
    job = league.get_job_info(pid)
    assert job['agent_num'] == 2 and len(job['player_id']) == 2
    league.finish_job({
        'launch_player': pid,
        'player_id': job['player_id'],
        'result': ['wins'] * 4,
    })
    league.update_active_player({'player_id': pid, 'train_iter': 8})  # 2x one_phase_step
    assert league.judge_snapshot(pid)


def test_metric_serial_evaluator():
    from ding.worker import MetricSerialEvaluator, IMetric

    class Acc(IMetric):

        def eval(self, inputs, label):
            return {'acc': float((inputs.argmax(dim=-1) == label).float().mean())}

        def reduce_mean(self, inputs):
            return {'acc': float(np.mean([x['acc'] for x in inputs]))}

        def gt(self, m1, m2):
            if m2 is None:
                return True
            v2 = m2['acc'] if isinstance(m2, dict) else m2
            return m1['acc'] >= v2

    class ArgmaxPolicy:

        def reset(self):
            pass

        def forward(self, inputs):
            return inputs

    data = [(torch.eye(4)[torch.randint(0, 4, (8, ))], torch.randint(0, 4, (8, ))) for _ in range(3)]
    # perfectly-predictable loader: logits one-hot == label
    perfect = [(torch.eye(4)[lbl], lbl) for _, lbl in data]
    ev = MetricSerialEvaluator(
        EasyDict({'type': 'metric', 'eval_freq': 1, 'stop_value': 0.99}),
        env=(perfect, Acc()), policy=ArgmaxPolicy(), exp_name='exp/test_metric_eval'
    )
    assert ev.should_eval(1)
    stop, result = ev.eval(None, train_iter=1)
    assert stop and result['acc'] == 1.0


def test_battle_interaction_evaluator():
    from ding.worker import BattleInteractionSerialEvaluator
    pols = [_FlattenBattlePolicy(p) for p in _two_dqn_policies()]
    env = _battle_env_manager()
    ev = BattleInteractionSerialEvaluator(
        EasyDict({'type': 'battle_interaction', 'n_episode': 3, 'stop_value': 1e9, 'eval_freq': 1}),
        env=env, policy=pols, exp_name='exp/test_battle_eval'
    )
    assert ev.should_eval(1)
    # DQN collect_mode needs eps; wrap to default it
    class _Eps:
        def __init__(self, p): self._p = p
        def forward(self, obs, **kw): return self._p.forward(obs, eps=0.1)
        def reset(self, *a, **k): return self._p.reset(*a, **k)
    ev.reset(_policy=[_Eps(p) for p in pols])
    stop, info = ev.eval(None, train_iter=1)
    assert not stop and len(info['eval_episode_return']) >= 3
    ev.close()


def test_pfsp_weightings():
    """PFSP math: harder opponents get more weight; degenerate cases uniform."""
    from ding.league.algorithm import pfsp
    wr = np.array([0.1, 0.5, 0.9])
    for weighting in ('variance', 'linear', 'linear_capped', 'squared'):
        p = pfsp(wr, weighting)
        assert abs(p.sum() - 1.0) < 1e-9
    sq = pfsp(wr, 'squared')
    assert sq[0] > sq[1] > sq[2], "squared PFSP must prefer opponents we lose to"
    var = pfsp(wr, 'variance')
    assert var[1] == max(var), "variance PFSP must prefer 50% opponents"
    # all-beaten: uniform fallback
    u = pfsp(np.array([1.0, 1.0]), 'squared')
    assert np.allclose(u, [0.5, 0.5])


def test_elo_rating_math():
    from ding.league.metric import EloCalculator
    a, b = EloCalculator.get_new_rating(1200, 1200, 1)
    assert a > 1200 > b and abs((a - 1200) + (b - 1200)) < 1e-6, "zero-sum update"
    # beating a much stronger player moves more points
    a2, _ = EloCalculator.get_new_rating(1200, 1600, 1)
    assert (a2 - 1200) > (a - 1200)
    # draws move ratings toward each other
    a3, b3 = EloCalculator.get_new_rating(1200, 1600, 0)
    assert a3 > 1200 and b3 < 1600


def test_trueskill_rating_math():
    from ding.league.metric import PlayerRating, TrueSkillCalculator
    a, b = PlayerRating(), PlayerRating()
    a2, b2 = TrueSkillCalculator.get_new_rating(a, b, 1)
    assert a2.mu > a.mu and b2.mu < b.mu
    assert a2.sigma < a.sigma, "uncertainty must shrink after a match"
    # exposure = mu - 3 sigma grows for the winner
    assert a2.exposure > a.exposure


def test_selfplay_demo_ppo_main_runs():
    """Reference dizoo/league_demo/selfplay_demo_ppo_main.py analog."""
    from dizoo.league_demo.selfplay_demo_ppo_main import main
    pols = main(max_train_iter=4)
    assert len(pols) == 2


def test_league_demo_ppo_main_snapshots():
    """Reference dizoo/league_demo/league_demo_ppo_main.py analog: the league
    loop must produce at least one historical snapshot."""
    from dizoo.league_demo.league_demo_ppo_main import main
    _, league, payoff = main(max_train_iter=25)
    assert len(league.historical_players) >= 1
