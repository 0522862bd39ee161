import pytest
import torch

from ding.model import (
    DQN, C51DQN, QRDQN, IQN, FQF, RainbowDQN, DRQN, BDQ, GTrXLDQN, VAC, ContinuousQAC, DiscreteQAC, model_wrap,
    create_model,
)

B, OBS, ACT = 4, 10, 5


def test_dqn_vector_and_image():
    m = DQN(OBS, ACT)
    out = m(torch.randn(B, OBS))
    assert out['logit'].shape == (B, ACT)
    m2 = DQN((4, 84, 84), ACT, encoder_hidden_size_list=[16, 16, 16, 32])
    out2 = m2(torch.randn(2, 4, 84, 84))
    assert out2['logit'].shape == (2, ACT)
    # multi-discrete
    m3 = DQN(OBS, [3, 4])
    out3 = m3(torch.randn(B, OBS))
    assert out3['logit'][0].shape == (B, 3) and out3['logit'][1].shape == (B, 4)


def test_distributional_models():
    c51 = C51DQN(OBS, ACT, n_atom=51)
    o = c51(torch.randn(B, OBS))
    assert o['logit'].shape == (B, ACT) and o['distribution'].shape == (B, ACT, 51)
    assert torch.allclose(o['distribution'].sum(-1), torch.ones(B, ACT), atol=1e-4)
    qr = QRDQN(OBS, ACT, num_quantiles=16)
    o = qr(torch.randn(B, OBS))
    assert o['q'].shape == (B, 16, ACT) and o['tau'].shape == (B, 16, 1)
    iqn = IQN(OBS, ACT, num_quantiles=8)
    o = iqn(torch.randn(B, OBS))
    assert o['q'].shape == (8, B, ACT) and o['quantiles'].shape == (8 * B, 1)
    fqf = FQF(OBS, ACT, num_quantiles=8)
    o = fqf(torch.randn(B, OBS))
    assert o['q'].shape == (B, 8, ACT) and o['quantiles'].shape == (B, 9)
    rb = RainbowDQN(OBS, ACT)
    o = rb(torch.randn(B, OBS))
    assert o['logit'].shape == (B, ACT)
    rb.reset_noise()


def test_drqn():
    T = 3
    m = DRQN(OBS, ACT)
    out = m({'obs': torch.randn(T, B, OBS), 'prev_state': None})
    assert out['logit'].shape == (T, B, ACT)
    assert len(out['next_state']) == B
    inf = m({'obs': torch.randn(B, OBS), 'prev_state': out['next_state']}, inference=True)
    assert inf['logit'].shape == (B, ACT)


def test_bdq():
    m = BDQ(OBS, num_branches=3, action_bins_per_branch=4)
    out = m(torch.randn(B, OBS))
    assert out['logit'].shape == (B, 3, 4)


def test_gtrxl_dqn():
    m = GTrXLDQN(OBS, ACT, memory_len=4, hidden_size=16)
    out = m(torch.randn(3, B, OBS))
    assert out['logit'].shape == (3, B, ACT)


def test_vac_modes():
    m = VAC(OBS, ACT)
    x = torch.randn(B, OBS)
    a = m(x, 'compute_actor')
    assert a['logit'].shape == (B, ACT)
    c = m(x, 'compute_critic')
    assert c['value'].shape == (B, )
    ac = m(x, 'compute_actor_critic')
    assert ac['logit'].shape == (B, ACT) and ac['value'].shape == (B, )
    mc = VAC(OBS, 3, action_space='continuous')
    out = mc(x, 'compute_actor')
    assert out['logit']['mu'].shape == (B, 3)


def test_qac():
    m = ContinuousQAC(OBS, 3, 'regression', twin_critic=True)
    x = torch.randn(B, OBS)
    a = m(x, 'compute_actor')
    assert a['action'].shape == (B, 3)
    q = m({'obs': x, 'action': a['action']}, 'compute_critic')
    assert len(q['q_value']) == 2 and q['q_value'][0].shape == (B, )
    ms = ContinuousQAC(OBS, 3, 'reparameterization')
    a = ms(x, 'compute_actor')
    assert a['logit'][0].shape == (B, 3)
    md = DiscreteQAC(OBS, ACT, twin_critic=True)
    a = md(x, 'compute_actor')
    assert a['logit'].shape == (B, ACT)
    q = md(x, 'compute_critic')
    assert q['q_value'][0].shape == (B, ACT)


def test_model_wrappers():
    m = DQN(OBS, ACT)
    wm = model_wrap(m, wrapper_name='eps_greedy_sample')
    out = wm.forward(torch.randn(B, OBS), eps=0.5)
    assert out['action'].shape == (B, )
    wm2 = model_wrap(m, wrapper_name='argmax_sample')
    out2 = wm2.forward(torch.randn(B, OBS))
    assert out2['action'].shape == (B, )

    target = model_wrap(DQN(OBS, ACT), wrapper_name='target', update_type='assign', update_kwargs={'freq': 2})
    target.update(m.state_dict(), direct=True)
    for p1, p2 in zip(m.parameters(), target.model.parameters()):
        assert torch.equal(p1, p2)
    tm = model_wrap(DQN(OBS, ACT), wrapper_name='target', update_type='momentum', update_kwargs={'theta': 0.5})
    before = next(tm.model.parameters()).clone()
    tm.update(m.state_dict())
    after = next(tm.model.parameters())
    assert not torch.equal(before, after)


def test_hidden_state_wrapper():
    m = DRQN(OBS, ACT)
    wm = model_wrap(m, wrapper_name='hidden_state', state_num=B)
    out = wm.forward(torch.randn(B, OBS), inference=True)
    assert 'next_state' in out
    wm.reset(data_id=[0, 1])
    out2 = wm.forward(torch.randn(B, OBS), inference=True)
    assert out2['logit'].shape == (B, ACT)


def test_create_model():
    cfg = dict(type='dqn', obs_shape=OBS, action_shape=ACT)
    m = create_model(cfg)
    assert isinstance(m, DQN)


def test_hpt_model():
    """HPT: perceiver policy stem + dueling head works as a DQN model."""
    import torch
    from ding.model.template import HPT
    m = HPT(4, 3)
    out = m(torch.randn(5, 4))
    assert out['logit'].shape == (5, 3)
    out['logit'].sum().backward()
    assert m.policy_stem.tokens.grad is not None
