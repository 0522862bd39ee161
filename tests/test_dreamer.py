"""DreamerV3 subsystem: RSSM, world-model training, latent-imagination
policy updates, sequence buffer, distributions."""
import numpy as np
import pytest
import torch

from ding.utils import EasyDict, deep_merge_dicts


def _wm_cfg():
    return EasyDict(dict(
        type='dreamer',
        import_names=['ding.world_model.dreamer'],
        train_freq=1, eval_freq=int(1e9), cuda=False,
        model=dict(
            state_size=4, obs_type='vector', action_size=2, action_type='discrete',
            encoder_hidden_size_list=[32, 32],
            dyn_stoch=8, dyn_deter=32, dyn_hidden=32, dyn_discrete=8,
            units=32, reward_layers=1, discount_layers=1, image_dec_layers=1,
            batch_size=4, batch_length=6,
        ),
    ))


def _fill_sequence_buffer(n=64):
    from ding.worker import create_buffer
    buf = create_buffer(EasyDict({'type': 'sequence', 'replay_buffer_size': 1000}))
    for i in range(n):
        buf.push({
            'obs': np.random.randn(4).astype(np.float32),
            'action': np.int64(i % 2),
            'reward': np.float32(np.random.randn()),
            'done': False,
        })
    return buf


def test_dreamer_world_model_train():
    from ding.world_model import create_world_model
    wm = create_world_model(_wm_cfg())
    buf = _fill_sequence_buffer()
    post, context = wm.train(buf, envstep=10, train_iter=0, batch_size=4, batch_length=6)
    assert post['logit'].shape == (4, 6, 8, 8)
    assert context['feat'].shape == (4, 6, 8 * 8 + 32)
    assert not post['stoch'].requires_grad
    # second step should run (params got an update)
    wm.train(buf, envstep=20, train_iter=1, batch_size=4, batch_length=6)


def test_dreamer_policy_learn_and_collect():
    from ding.world_model import create_world_model
    from ding.policy import create_policy
    wm = create_world_model(_wm_cfg())
    buf = _fill_sequence_buffer()
    post, _ = wm.train(buf, envstep=10, train_iter=0, batch_size=4, batch_length=6)

    cfg = EasyDict(deep_merge_dicts(
        __import__('ding.policy.dreamer', fromlist=['DREAMERPolicy']).DREAMERPolicy.default_config(),
        EasyDict(dict(
            type='dreamer', cuda=False, imag_horizon=5,
            model=dict(action_shape=2, dyn_stoch=8, dyn_deter=32, dyn_discrete=8,
                       units=32, actor_layers=1, value_layers=1, actor_dist='onehot'),
            learn=dict(batch_size=4, batch_length=6),
            collect=dict(unroll_len=1, action_size=2, collect_dyn_sample=True),
        ))
    ))
    pol = create_policy(cfg, enable_field=['learn', 'collect', 'eval'])
    info = pol._forward_learn(post, world_model=wm, envstep=10)
    assert np.isfinite(info['actor_loss']) and np.isfinite(info['critic_loss'])

    # collect: first step (no state), then a step carrying state + reset
    obs = {0: torch.randn(4), 1: torch.randn(4)}
    out = pol._forward_collect(obs, world_model=wm, envstep=10)
    assert set(out.keys()) == {0, 1}
    assert out[0]['action'].dtype in (torch.int64, torch.long)
    state = [out[i]['state'] for i in range(2)]
    out2 = pol._forward_collect(obs, world_model=wm, envstep=11, reset=np.array([1, 0]), state=state)
    assert 'action' in out2[0]
    ev = pol._forward_eval(obs, world_model=wm)
    assert 'action' in ev[0]


def test_dreamer_distributions():
    from ding.torch_utils.network.dreamer import TwoHotDistSymlog, OneHotDist, SymlogDist, symlog, symexp
    # symlog/symexp roundtrip
    x = torch.randn(32) * 10
    assert torch.allclose(symexp(symlog(x)), x, atol=1e-4, rtol=1e-4)
    # twohot: log_prob maximal near the encoded value
    logits = torch.zeros(1, 255, requires_grad=True)
    d = TwoHotDistSymlog(logits)
    lp = d.log_prob(torch.tensor([[3.0]]))
    lp.sum().backward()
    assert logits.grad is not None
    assert d.mean().shape == (1, 1)
    # onehot unimix: probs bounded away from 0
    od = OneHotDist(torch.tensor([[10.0, -10.0]]), unimix_ratio=0.1)
    assert od.probs.min() >= 0.04
    s = od.sample()
    assert s.shape == (1, 2)
    # symlog dist log_prob is -distance
    sd = SymlogDist(torch.zeros(2, 3), dim_to_reduce=[-1])
    lp = sd.log_prob(torch.zeros(2, 3))
    assert torch.allclose(lp, torch.zeros(2))


def test_dreamer_pipeline_smoke():
    """serial_pipeline_dreamer end-to-end on cartpole for 2 iterations."""
    from ding.entry import serial_pipeline_dreamer
    main = EasyDict(dict(
        exp_name='exp/test_dreamer',
        env=dict(collector_env_num=2, evaluator_env_num=2, n_evaluator_episode=2, stop_value=195),
        policy=dict(
            cuda=False, random_collect_size=24, imag_horizon=4,
            model=dict(action_shape=2, dyn_stoch=8, dyn_deter=32, dyn_discrete=8,
                       units=32, actor_layers=1, value_layers=1, actor_dist='onehot'),
            learn=dict(batch_size=4, batch_length=6, learning_rate=3e-4),
            collect=dict(n_sample=16, unroll_len=1, action_size=2, collect_dyn_sample=True),
            eval=dict(evaluator=dict(eval_freq=int(1e6))),
            other=dict(replay_buffer=dict(type='sequence', replay_buffer_size=1000)),
        ),
        world_model=dict(
            type='dreamer', import_names=['ding.world_model.dreamer'],
            pretrain=1, train_freq=1, cuda=False,
            model=dict(
                state_size=4, obs_type='vector', action_size=2, action_type='discrete',
                encoder_hidden_size_list=[32, 32], dyn_stoch=8, dyn_deter=32, dyn_hidden=32,
                dyn_discrete=8, units=32, reward_layers=1, discount_layers=1, image_dec_layers=1,
                batch_size=4, batch_length=6,
            ),
        ),
    ))
    create = EasyDict(dict(
        env=dict(type='cartpole', import_names=['dizoo.classic_control.cartpole.envs.cartpole_env']),
        env_manager=dict(type='base'),
        policy=dict(type='dreamer'),
        world_model=dict(type='dreamer', import_names=['ding.world_model.dreamer']),
    ))
    serial_pipeline_dreamer((main, create), seed=0, max_train_iter=2)
