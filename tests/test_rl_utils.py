"""Numerics tests for the pure-PyTorch rl_utils lane (the oracle the HIP
kernels are verified against in tests/test_ops_gpu.py)."""
import numpy as np
import pytest
import torch

from ding.rl_utils import (
    gae, gae_data, q_nstep_td_data, q_nstep_td_error, q_nstep_td_error_with_rescale, q_1step_td_data,
    q_1step_td_error, dist_nstep_td_data, dist_nstep_td_error, td_lambda_data, td_lambda_error,
    generalized_lambda_returns, v_nstep_td_data, v_nstep_td_error, ppo_data, ppo_error, ppo_error_continuous,
    ppo_data_continuous, vtrace_data, vtrace_error_discrete_action, upgo_loss, compute_q_retraces,
    qrdqn_nstep_td_data, qrdqn_nstep_td_error, iqn_nstep_td_data, iqn_nstep_td_error, q_nstep_sql_td_error,
    get_epsilon_greedy_fn, Adder, value_transform, value_inv_transform, a2c_data, a2c_error, coma_data, coma_error,
    nstep_return, nstep_return_data, m_q_1step_td_data, m_q_1step_td_error, dqfd_nstep_td_data, dqfd_nstep_td_error,
)

T, B, N = 6, 4, 5


def test_gae_matches_manual():
    torch.manual_seed(0)
    value = torch.randn(T, B)
    next_value = torch.randn(T, B)
    reward = torch.randn(T, B)
    adv = gae(gae_data(value, next_value, reward, None, None), 0.99, 0.95)
    # manual per-column reverse recursion
    delta = reward + 0.99 * next_value - value
    expect = torch.zeros_like(delta)
    acc = torch.zeros(B)
    for t in reversed(range(T)):
        acc = delta[t] + 0.99 * 0.95 * acc
        expect[t] = acc
    assert torch.allclose(adv, expect, atol=1e-6)


def test_gae_done_resets_scan():
    value = torch.zeros(3, 1)
    next_value = torch.ones(3, 1)
    reward = torch.ones(3, 1)
    done = torch.tensor([[0.], [1.], [0.]])
    adv = gae(gae_data(value, next_value, reward, done, None), 1.0, 1.0)
    # t=1 done: next_value masked, scan restarts above
    assert adv[2, 0] == 2.0  # 1 + 1*1
    assert adv[1, 0] == 1.0  # reward only: next_value masked, scan cut at t=1
    # delta0 = 1 + 1 - 0 = 2, factor0 = 1*(1-traj_flag0)=1 => adv0 = 2 + adv1 = 3
    assert adv[0, 0] == 3.0


def test_q_nstep_td():
    torch.manual_seed(1)
    nstep = 3
    q = torch.randn(B, N, requires_grad=True)
    next_n_q = torch.randn(B, N)
    action = torch.randint(0, N, (B, ))
    next_action = torch.randint(0, N, (B, ))
    reward = torch.randn(nstep, B)
    done = torch.zeros(B)
    data = q_nstep_td_data(q, next_n_q, action, next_action, reward, done, None)
    loss, td = q_nstep_td_error(data, 0.95, nstep=nstep)
    assert loss.shape == () and td.shape == (B, )
    loss.backward()
    assert q.grad is not None
    # manual target for sample 0
    g = 0.95
    target = reward[0, 0] + g * reward[1, 0] + g ** 2 * reward[2, 0] + g ** 3 * next_n_q[0, next_action[0]]
    assert torch.allclose(td[0], (q.detach()[0, action[0]] - target) ** 2, atol=1e-5)


def test_q_nstep_td_rescale_inverts():
    x = torch.randn(10) * 5
    assert torch.allclose(value_inv_transform(value_transform(x)), x, atol=1e-4)
    nstep = 2
    q = torch.randn(B, N, requires_grad=True)
    data = q_nstep_td_data(
        q, torch.randn(B, N), torch.randint(0, N, (B, )), torch.randint(0, N, (B, )), torch.randn(nstep, B),
        torch.zeros(B), None
    )
    loss, td = q_nstep_td_error_with_rescale(data, 0.99, nstep=nstep)
    loss.backward()


def test_dist_nstep_td():
    torch.manual_seed(2)
    n_atom, nstep = 51, 3
    dist = torch.softmax(torch.randn(B, N, n_atom), -1).requires_grad_(True)
    next_dist = torch.softmax(torch.randn(B, N, n_atom), -1)
    act = torch.randint(0, N, (B, ))
    next_act = torch.randint(0, N, (B, ))
    reward = torch.randn(nstep, B)
    done = torch.zeros(B)
    data = dist_nstep_td_data(dist, next_dist, act, next_act, reward, done, None)
    loss, td = dist_nstep_td_error(data, 0.99, -10, 10, n_atom, nstep)
    assert td.shape == (B, )
    loss.backward()
    assert dist.grad is not None


def test_td_lambda_and_generalized_returns():
    torch.manual_seed(3)
    value = torch.randn(T + 1, B, requires_grad=True)
    reward = torch.randn(T, B)
    loss = td_lambda_error(td_lambda_data(value, reward, None), 0.9, 0.8)
    loss.backward()
    # lambda=1 equals discounted MC return + terminal bootstrap
    v = torch.randn(T + 1, B)
    r = torch.randn(T, B)
    ret = generalized_lambda_returns(v, r, 0.9, 1.0)
    expect = torch.zeros(T, B)
    acc = v[-1]
    for t in reversed(range(T)):
        acc = r[t] + 0.9 * acc
        expect[t] = acc
    assert torch.allclose(ret, expect, atol=1e-5)


def test_ppo_error():
    torch.manual_seed(4)
    data = ppo_data(
        torch.randn(B, N).requires_grad_(True), torch.randn(B, N), torch.randint(0, N, (B, )),
        torch.randn(B).requires_grad_(True), torch.randn(B), torch.randn(B), torch.randn(B), None
    )
    loss, info = ppo_error(data)
    total = loss.policy_loss + 0.5 * loss.value_loss - 0.01 * loss.entropy_loss
    total.backward()
    assert np.isscalar(info.approx_kl) and 0 <= info.clipfrac <= 1
    # dual clip path
    loss2, _ = ppo_error(
        ppo_data(
            torch.randn(B, N), torch.randn(B, N), torch.randint(0, N, (B, )), torch.randn(B), torch.randn(B),
            torch.randn(B), torch.randn(B), None
        ),
        dual_clip=5.0
    )


def test_ppo_error_continuous():
    torch.manual_seed(5)
    D = 3
    mu_new = torch.randn(B, D).requires_grad_(True)
    data = ppo_data_continuous(
        {'mu': mu_new, 'sigma': torch.rand(B, D) + 0.5},
        {'mu': torch.randn(B, D), 'sigma': torch.rand(B, D) + 0.5}, torch.randn(B, D),
        torch.randn(B).requires_grad_(True), torch.randn(B), torch.randn(B), torch.randn(B), None
    )
    loss, info = ppo_error_continuous(data)
    (loss.policy_loss + loss.value_loss).backward()
    assert mu_new.grad is not None


def test_vtrace():
    torch.manual_seed(6)
    value = torch.randn(T + 1, B, requires_grad=True)
    target = torch.randn(T, B, N, requires_grad=True)
    behaviour = torch.randn(T, B, N)
    action = torch.randint(0, N, (T, B))
    reward = torch.rand(T, B)
    loss = vtrace_error_discrete_action(vtrace_data(target, behaviour, action, value, reward, None))
    (loss.policy_loss + loss.value_loss - 0.01 * loss.entropy_loss).backward()
    assert value.grad is not None and target.grad is not None
    # on-policy (target == behaviour, rho=c=1, lambda=1) -> vs == lambda returns
    from ding.rl_utils.vtrace import vtrace_nstep_return
    rhos = torch.ones(T, B)
    vs = vtrace_nstep_return(rhos, rhos, reward, value.detach(), gamma=0.9, lambda_=1.0)
    expect = generalized_lambda_returns(value.detach(), reward, 0.9, 1.0)
    assert torch.allclose(vs, expect, atol=1e-4)


def test_upgo():
    torch.manual_seed(7)
    target = torch.randn(T, B, N, requires_grad=True)
    rhos = torch.rand(T, B)
    action = torch.randint(0, N, (T, B))
    rewards = torch.randn(T, B)
    bootstrap = torch.randn(T + 1, B)
    loss = upgo_loss(target, rhos, action, rewards, bootstrap)
    loss.backward()


def test_retrace():
    torch.manual_seed(8)
    q_values = torch.randn(T + 1, B, N)
    v_pred = torch.randn(T + 1, B, 1)
    rewards = torch.randn(T, B)
    actions = torch.randint(0, N, (T, B))
    weights = torch.ones(T, B)
    ratio = torch.rand(T, B, N) * 2
    out = compute_q_retraces(q_values, v_pred, rewards, actions, weights, ratio, 0.99)
    assert out.shape == (T + 1, B, 1)
    assert torch.allclose(out[-1], v_pred[-1])


def test_quantile_family():
    torch.manual_seed(9)
    nstep, tau = 3, 8
    q = torch.randn(B, tau, N, requires_grad=True)
    data = qrdqn_nstep_td_data(
        q, torch.randn(B, tau, N), torch.randint(0, N, (B, )), torch.randint(0, N, (B, )), torch.randn(nstep, B),
        torch.zeros(B), tau, None
    )
    loss, td = qrdqn_nstep_td_error(data, 0.99, nstep)
    loss.backward()
    q2 = torch.randn(tau, B, N, requires_grad=True)
    data2 = iqn_nstep_td_data(
        q2, torch.randn(tau, B, N), torch.randint(0, N, (B, )), torch.randint(0, N, (B, )), torch.randn(nstep, B),
        torch.zeros(B), torch.rand(tau, B, 1), None
    )
    loss2, td2 = iqn_nstep_td_error(data2, 0.99, nstep)
    loss2.backward()


def test_sql_and_munchausen():
    q = torch.randn(B, N, requires_grad=True)
    data = q_nstep_td_data(
        q, torch.randn(B, N), torch.randint(0, N, (B, )), torch.randint(0, N, (B, )), torch.randn(2, B),
        torch.zeros(B), None
    )
    loss, td = q_nstep_sql_td_error(data, 0.99, alpha=0.1, nstep=2)
    loss.backward()
    q3 = torch.randn(B, N, requires_grad=True)
    m_data = m_q_1step_td_data(q3, torch.randn(B, N), torch.randn(B, N), torch.randint(0, N, (B, )), torch.randn(B),
                               torch.zeros(B), None)
    loss3, td3, qmean = m_q_1step_td_error(m_data, 0.99, 0.03, 0.9)
    loss3.backward()


def test_dqfd():
    nstep = 3
    q = torch.randn(B, N, requires_grad=True)
    data = dqfd_nstep_td_data(
        q, torch.randn(B, N), torch.randint(0, N, (B, )), torch.randint(0, N, (B, )), torch.randn(nstep, B),
        torch.zeros(B), torch.zeros(B), None, torch.randn(B, N), torch.randint(0, N, (B, )),
        torch.randint(0, 2, (B, ))
    )
    loss, per_sample, parts = dqfd_nstep_td_error(data, 0.99, 1.0, 1.0, 0.8, nstep=nstep)
    loss.backward()


def test_a2c_coma():
    data = a2c_data(
        torch.randn(B, N, requires_grad=True), torch.randint(0, N, (B, )), torch.randn(B, requires_grad=True),
        torch.randn(B), torch.randn(B), None
    )
    loss = a2c_error(data)
    (loss.policy_loss + loss.value_loss).backward()
    A = 3
    coma = coma_data(
        torch.randn(T, B, A, N, requires_grad=True), torch.randint(0, N, (T, B, A)),
        torch.randn(T, B, A, N, requires_grad=True), torch.randn(T, B, A, N), torch.randn(T, B), None
    )
    closs = coma_error(coma, 0.99, 0.8)
    (closs.policy_loss + closs.q_value_loss).backward()


def test_exploration_and_adder():
    eps_fn = get_epsilon_greedy_fn(0.95, 0.05, 10000, 'exp')
    assert abs(eps_fn(0) - 0.95) < 1e-6 and eps_fn(10 ** 7) == pytest.approx(0.05, abs=1e-3)
    lin = get_epsilon_greedy_fn(1.0, 0.1, 100, 'linear')
    assert lin(50) == pytest.approx(0.55) and lin(1000) == 0.1

    data = [
        {'obs': torch.randn(4), 'value': torch.randn(1), 'reward': torch.randn(1), 'done': False, 'action': torch.tensor([0])}
        for _ in range(10)
    ]
    out = Adder.get_gae(data, torch.zeros(1), 0.99, 0.95, cuda=False)
    assert all('adv' in d for d in out)
    from collections import deque
    nd = Adder.get_nstep_return_data(deque(data), nstep=3, gamma=0.99)
    assert list(nd)[0]['reward'].shape[-1] == 3
    samples = Adder.get_train_sample(data, unroll_len=4, last_fn_type='last')
    assert len(samples) == 3


def test_nstep_return_value_gamma():
    nstep = 4
    reward = torch.randn(nstep, B)
    next_v = torch.randn(B)
    done = torch.zeros(B)
    base = nstep_return(nstep_return_data(reward, next_v, done), 0.9, nstep)
    vg = nstep_return(nstep_return_data(reward, next_v, done), 0.9, nstep, value_gamma=torch.full((B, ), 0.9 ** nstep))
    assert torch.allclose(base, vg, atol=1e-6)
