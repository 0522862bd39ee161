"""Plan-diffuser (trajectory diffusion) tests."""
import numpy as np
import pytest
import torch

from ding.utils import EasyDict


HORIZON, OBS, ACT = 8, 4, 2


def _pd_model(value=True):
    from ding.model.template.diffusion import PlanDiffuser
    return PlanDiffuser(
        diffuser_model='GaussianDiffusion',
        diffuser_model_cfg=dict(
            model='DiffusionUNet1d',
            model_cfg=dict(transition_dim=OBS + ACT, dim=16, dim_mults=[1, 2]),
            horizon=HORIZON, obs_dim=OBS, action_dim=ACT, n_timesteps=8, clip_denoised=True,
        ),
        value_model='ValueDiffusion' if value else None,
        value_model_cfg=dict(
            model='TemporalValue',
            model_cfg=dict(horizon=HORIZON, transition_dim=OBS + ACT, dim=16, dim_mults=[1, 2]),
            horizon=HORIZON, obs_dim=OBS, action_dim=ACT, n_timesteps=8,
        ) if value else None,
        scale=0.1, t_stopgrad=2, n_guide_steps=1,
    )


def test_diffusion_losses_and_sample():
    m = _pd_model()
    B = 6
    x = torch.randn(B, HORIZON, OBS + ACT)
    cond = {0: torch.randn(B, OBS)}
    t = torch.randint(0, 8, (B, ))
    dloss, a0 = m.diffuser_loss(x, cond, t)
    assert dloss.requires_grad
    vloss, logs = m.value_loss(x, cond, torch.randn(B, 1), t)
    assert vloss.requires_grad and 'mean_pred' in logs
    act = m.get_eval({0: torch.randn(2, OBS)}, batch_size=3)
    assert act.shape == (2, ACT)


def test_pd_policy_learn_eval():
    from ding.policy import create_policy
    from ding.policy.plan_diffuser import PDPolicy
    from ding.utils import deep_merge_dicts
    cfg = EasyDict(dict(
        type='pd', cuda=False, on_policy=False, priority=False, priority_IS_weight=False,
        model=dict(
            type='pd', import_names=['ding.model.template.diffusion'],
            diffuser_model='GaussianDiffusion',
            diffuser_model_cfg=dict(
                model='DiffusionUNet1d',
                model_cfg=dict(transition_dim=OBS + ACT, dim=16, dim_mults=[1, 2]),
                horizon=HORIZON, obs_dim=OBS, action_dim=ACT, n_timesteps=8, clip_denoised=True,
            ),
            value_model='ValueDiffusion',
            value_model_cfg=dict(
                model='TemporalValue',
                model_cfg=dict(horizon=HORIZON, transition_dim=OBS + ACT, dim=16, dim_mults=[1, 2]),
                horizon=HORIZON, obs_dim=OBS, action_dim=ACT, n_timesteps=8,
            ),
            scale=0.1, t_stopgrad=2, n_guide_steps=1,
        ),
        learn=dict(
            batch_size=4, learning_rate=1e-3, gradient_accumulate_every=1, plan_batch_size=2,
            update_target_freq=5, step_start_update_target=10, target_weight=0.99, value_step=100,
            include_returns=True, discount_factor=0.99, ignore_done=False,
        ),
        collect=dict(unroll_len=1),
        eval=dict(evaluator=dict(eval_freq=1000)),
        other=dict(replay_buffer=dict(replay_buffer_size=100)),
    ))
    cfg = EasyDict(deep_merge_dicts(PDPolicy.default_config(), cfg))
    pol = create_policy(cfg, enable_field=['learn', 'eval'])
    batch = [
        {
            'trajectories': torch.randn(HORIZON, OBS + ACT),
            'condition_id': [0],
            'condition_val': torch.randn(OBS),
            'returns': torch.randn(1),
            'obs': torch.randn(OBS), 'action': torch.randn(ACT), 'reward': torch.randn(1), 'done': False,
        } for _ in range(4)
    ]
    info = pol._forward_learn(batch)
    assert np.isfinite(info['diffuse_loss'])
    out = pol._forward_eval({0: torch.randn(OBS), 1: torch.randn(OBS)})
    assert out[0]['action'].shape == (ACT, )


def test_decision_diffuser_dd():
    from ding.model.template.diffusion import GaussianInvDynDiffusion
    m = GaussianInvDynDiffusion(
        model='DiffusionUNet1d',
        model_cfg=dict(transition_dim=OBS, dim=16, dim_mults=[1, 2], returns_condition=True),
        horizon=HORIZON, obs_dim=OBS, action_dim=ACT, n_timesteps=8, hidden_dim=32,
        returns_condition=True,
    )
    B = 5
    x = torch.randn(B, HORIZON, OBS)
    cond = {0: torch.randn(B, OBS)}
    t = torch.randint(0, 8, (B, ))
    loss = m.p_losses(x, cond, t, returns=torch.randn(B, 1))
    assert loss.requires_grad
    # inverse dynamics
    pair = torch.cat([x[:, 0], x[:, 1]], dim=-1)
    act = m.inv_model(pair)
    assert act.shape == (B, ACT)
    # conditioned sampling with classifier-free guidance
    sample = m.conditional_sample({0: torch.randn(2, OBS)}, returns=torch.randn(2, 1))
    assert sample.trajectories.shape == (2, HORIZON, OBS)


def test_qgpo_model_and_policy():
    from ding.model.template.qgpo import QGPO, marginal_prob_std, dpm_solver_sample
    from ding.policy import create_policy
    from ding.policy.qgpo import QGPOPolicy
    from ding.utils import deep_merge_dicts
    obs_dim, act_dim, B, M = 4, 2, 8, 4
    cfg = EasyDict(dict(obs_dim=obs_dim, action_dim=act_dim, qgpo_critic=dict(alpha=3, q_alpha=1)))
    m = QGPO(cfg)
    # marginal prob std sane: alpha->1, std->0 at t->0
    a0, s0 = marginal_prob_std(torch.tensor([1e-4]))
    assert float(a0) > 0.99 and float(s0) < 0.05
    # sampling produces finite actions
    states = np.random.randn(3, obs_dim).astype(np.float32)
    acts = m.select_actions(states, diffusion_steps=4)
    assert len(acts) == 3 and acts[0].shape == (act_dim, )
    support = m.sample(states, sample_per_state=M, diffusion_steps=3)
    assert support.shape == (3, M, act_dim)
    # policy learn phases
    pcfg = EasyDict(deep_merge_dicts(QGPOPolicy.default_config(), EasyDict(dict(
        type='qgpo', cuda=False,
        model=dict(type='qgpo', import_names=['ding.model.template.qgpo'],
                   obs_dim=obs_dim, action_dim=act_dim, qgpo_critic=dict(alpha=3, q_alpha=1)),
        learn=dict(learning_rate=1e-4, batch_size=B, behavior_policy_stop_training_iter=1,
                   energy_guided_policy_begin_training_iter=1, q_value_stop_training_iter=3),
        eval=dict(guidance_scale=[0.0, 1.0], diffusion_steps=3),
    ))))
    pol = create_policy(pcfg, enable_field=['learn', 'eval'])
    data = dict(
        s=torch.randn(B, obs_dim), a=torch.randn(B, act_dim), r=torch.randn(B, 1),
        s_=torch.randn(B, obs_dim), d=torch.zeros(B, 1),
        fake_a=torch.randn(B, M, act_dim), fake_a_=torch.randn(B, M, act_dim),
    )
    info1 = pol._forward_learn(data)  # behavior phase
    assert info1['behavior_model_training_loss'] > 0
    info2 = pol._forward_learn(data)  # energy phase (q0 + qt)
    assert info2['q0_loss'] != 0 and info2['qt_loss'] != 0
    out = pol._forward_eval({0: torch.randn(obs_dim), 1: torch.randn(obs_dim)}, guidance_scale=1.0)
    assert out[0]['action'].shape == (act_dim, )
