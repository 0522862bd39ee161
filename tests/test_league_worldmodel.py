"""League + world-model subsystem tests."""
import numpy as np
import pytest
import torch

from ding.league import (
    BaseLeague, OneVsOneLeague, pfsp, EloCalculator, TrueSkillCalculator, PlayerRating, BattleSharedPayoff,
    HistoricalPlayer, create_league,
)
from ding.utils import EasyDict


def test_pfsp():
    w = np.array([0.9, 0.5, 0.1])
    p = pfsp(w, 'squared')
    assert p.argmax() == 2 and abs(p.sum() - 1) < 1e-9
    assert pfsp(np.ones(3), 'squared').sum() == pytest.approx(1)


def test_elo_trueskill():
    a, b = EloCalculator.get_new_rating(1200, 1200, 1)
    assert a > 1200 > b
    ra, rb = PlayerRating(), PlayerRating()
    ra2, rb2 = TrueSkillCalculator.get_new_rating(ra, rb, 1)
    assert ra2.mu > ra.mu and rb2.mu < rb.mu
    assert ra2.sigma < ra.sigma


def test_league_lifecycle(tmp_path):
    cfg = BaseLeague.default_config()
    cfg.path_policy = str(tmp_path / 'league')
    cfg.main_player.one_phase_step = 10
    league = BaseLeague(cfg)
    assert len(league.active_players) == 1
    pid = league.active_players[0].player_id
    job = league.get_job_info(pid)
    assert job['launch_player'] == pid
    league.update_active_player({'player_id': pid, 'train_iter': 100})
    snap = league.judge_snapshot(pid)
    assert snap and len(league.historical_players) == 1
    league.finish_job({
        'player_id': [pid, league.historical_players[0].player_id],
        'result': [['wins', 'wins', 'losses']],
    })
    # payoff visible
    wr = league.payoff[league.active_players[0], league.historical_players[0]]
    assert 0 <= wr <= 1


def test_payoff_decay():
    payoff = BattleSharedPayoff(EasyDict(dict(decay=0.9, min_win_rate_games=1)))

    class P:
        def __init__(self, pid):
            self.player_id = pid

    a, b = P('a'), P('b')
    payoff.add_player(a)
    payoff.add_player(b)
    payoff.update({'player_id': ['a', 'b'], 'result': [['wins'] * 8]})
    assert payoff[a, b] > 0.9


def test_mbpo_world_model():
    from ding.world_model import MBPOWorldModel
    from ding.worker import NaiveReplayBuffer
    cfg = MBPOWorldModel.default_config()
    cfg.model.state_size = 3
    cfg.model.action_size = 1
    cfg.model.hidden_size = 32
    cfg.model.ensemble_size = 3
    cfg.model.elite_size = 2
    wm = MBPOWorldModel(cfg)
    buf = NaiveReplayBuffer(NaiveReplayBuffer.default_config())
    for _ in range(64):
        buf.push({
            'obs': torch.randn(3), 'action': torch.randn(1), 'reward': torch.tensor([0.5]),
            'next_obs': torch.randn(3), 'done': False
        })
    wm.train(buf, envstep=0, train_iter=0)
    reward, next_obs, done = wm.step(torch.randn(5, 3), torch.randn(5, 1))
    assert reward.shape == (5, ) and next_obs.shape == (5, 3) and done.shape == (5, )
    assert wm.should_train(envstep=10000)


def test_ddppo_grad_flow():
    from ding.world_model import DDPPOWorldMode
    cfg = DDPPOWorldMode.default_config()
    cfg.model.state_size = 3
    cfg.model.action_size = 1
    cfg.model.hidden_size = 32
    cfg.model.ensemble_size = 3
    cfg.model.elite_size = 2
    wm = DDPPOWorldMode(cfg)
    obs = torch.randn(4, 3)
    action = torch.randn(4, 1, requires_grad=True)
    reward, next_obs, done = wm.step(obs, action)
    reward.sum().backward()
    assert action.grad is not None


def test_inverse_dynamics_model_heads():
    """IDM predicts a_t from (s_t, s_{t+1}); all three heads + training.
    Parity: reference ding/world_model/idm.py."""
    import torch
    from ding.world_model import InverseDynamicsModel
    x = torch.randn(8, 8)  # concatenated (s, s') pairs, obs_dim 4
    m = InverseDynamicsModel(4, 3, [32, 32], action_space='discrete')
    assert m.forward(x)['logit'].shape == (8, 3)
    assert m.predict_action(x)['action'].shape == (8, )
    y = torch.randint(0, 3, (8, ))
    l0 = m.train({'obs': x, 'action': y}, n_epoch=1, learning_rate=1e-2)
    l1 = m.train({'obs': x, 'action': y}, n_epoch=30, learning_rate=1e-2)
    assert l1 < l0, "IDM training must reduce the loss on a fixed batch"
    m2 = InverseDynamicsModel(4, 2, [32, 32], action_space='regression')
    assert m2.forward(x)['action'].shape == (8, 2)
    m3 = InverseDynamicsModel(4, 2, [32, 32], action_space='reparameterization')
    out = m3.forward(x)
    assert out['action'].shape == (8, 2) and out['action'].abs().max() <= 1.0
    m4 = InverseDynamicsModel([3, 36, 36], 5, [16, 16, 32], action_space='discrete')
    assert m4.forward(torch.randn(4, 6, 36, 36))['logit'].shape == (4, 5)
