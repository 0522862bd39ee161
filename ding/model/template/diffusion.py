"""Trajectory diffusion models: Diffuser (plan with value guidance) and
Decision Diffuser (obs-only diffusion + inverse-dynamics action extraction).

Parity: reference ding/model/template/diffusion.py (GaussianDiffusion:73,
ValueDiffusion:296, PlanDiffuser 'pd':321, GaussianInvDynDiffusion 'dd':372,
default_sample_fn:15, n_step_guided_p_sample:37).
"""
from collections import namedtuple
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ding.torch_utils.network.diffusion import (
    DiffusionUNet1d, TemporalValue, apply_conditioning, cosine_beta_schedule, extract,
)
from ding.utils import MODEL_REGISTRY

Sample = namedtuple('Sample', 'trajectories values chains')

_BACKBONES = {'DiffusionUNet1d': DiffusionUNet1d, 'TemporalValue': TemporalValue}


def _resolve(model):
    if isinstance(model, str):
        return _BACKBONES[model]
    return model


def default_sample_fn(model, x, cond, t):
    b = x.shape[0]
    model_mean, _, model_log_variance = model.p_mean_variance(x=x, cond=cond, t=t)
    noise = 0.5 * torch.randn_like(x)
    nonzero_mask = (1 - (t == 0).float()).reshape(b, *((1, ) * (len(x.shape) - 1)))
    values = torch.zeros(len(x), device=x.device)
    return model_mean + nonzero_mask * (0.5 * model_log_variance).exp() * noise, values


def get_guide_output(guide, x, cond, t):
    x.requires_grad_()
    y = guide(x, cond, t).squeeze(dim=-1)
    grad = torch.autograd.grad([y.sum()], [x])[0]
    x.detach()
    return y, grad


def n_step_guided_p_sample(model, x, cond, t, guide, scale: float = 0.001, t_stopgrad: int = 0,
                           n_guide_steps: int = 1, scale_grad_by_std: bool = True):
    """Guide the reverse step by ascending the value model's gradient."""
    model_log_variance = extract(model.posterior_log_variance_clipped, t, x.shape)
    model_std = torch.exp(0.5 * model_log_variance)
    model_var = torch.exp(model_log_variance)
    for _ in range(n_guide_steps):
        with torch.enable_grad():
            y, grad = get_guide_output(guide, x, cond, t)
        if scale_grad_by_std:
            grad = model_var * grad
        grad[t < t_stopgrad] = 0
        x = x + scale * grad
        x = apply_conditioning(x, cond, model.action_dim)
    model_mean, _, model_log_variance = model.p_mean_variance(x=x, cond=cond, t=t)
    noise = torch.randn_like(x)
    noise[t == 0] = 0
    return model_mean + model_std * noise, y


class GaussianDiffusion(nn.Module):
    """DDPM over [B, horizon, action+obs] trajectories."""

    def __init__(
        self,
        model: str,
        model_cfg: dict,
        horizon: int,
        obs_dim: int,
        action_dim: int,
        n_timesteps: int = 1000,
        predict_epsilon: bool = True,
        loss_discount: float = 1.0,
        clip_denoised: bool = False,
        action_weight: float = 1.0,
        loss_weights: Optional[dict] = None,
    ):
        super().__init__()
        self.horizon = horizon
        self.obs_dim = obs_dim
        self.action_dim = action_dim
        self.transition_dim = obs_dim + action_dim
        self.model = _resolve(model)(**model_cfg)
        self.predict_epsilon = predict_epsilon
        self.clip_denoised = clip_denoised
        self.n_timesteps = int(n_timesteps)
        self._register_schedule(cosine_beta_schedule(n_timesteps))
        self.loss_weights = self.get_loss_weights(action_weight, loss_discount, loss_weights)

    def _register_schedule(self, betas: torch.Tensor) -> None:
        alphas = 1. - betas
        alphas_cumprod = torch.cumprod(alphas, dim=0)
        alphas_cumprod_prev = torch.cat([torch.ones(1), alphas_cumprod[:-1]])
        self.register_buffer('betas', betas)
        self.register_buffer('alphas_cumprod', alphas_cumprod)
        self.register_buffer('alphas_cumprod_prev', alphas_cumprod_prev)
        self.register_buffer('sqrt_alphas_cumprod', torch.sqrt(alphas_cumprod))
        self.register_buffer('sqrt_one_minus_alphas_cumprod', torch.sqrt(1. - alphas_cumprod))
        self.register_buffer('sqrt_recip_alphas_cumprod', torch.sqrt(1. / alphas_cumprod))
        self.register_buffer('sqrt_recipm1_alphas_cumprod', torch.sqrt(1. / alphas_cumprod - 1))
        posterior_variance = betas * (1. - alphas_cumprod_prev) / (1. - alphas_cumprod)
        self.register_buffer('posterior_variance', posterior_variance)
        self.register_buffer(
            'posterior_log_variance_clipped', torch.log(torch.clamp(posterior_variance, min=1e-20))
        )
        self.register_buffer(
            'posterior_mean_coef1', betas * torch.sqrt(alphas_cumprod_prev) / (1. - alphas_cumprod)
        )
        self.register_buffer(
            'posterior_mean_coef2', (1. - alphas_cumprod_prev) * torch.sqrt(alphas) / (1. - alphas_cumprod)
        )

    def get_loss_weights(self, action_weight: float, discount: float, weights_dict: Optional[dict]):
        self.action_weight = action_weight
        dim_weights = torch.ones(self.transition_dim, dtype=torch.float32)
        for ind, w in (weights_dict or {}).items():
            dim_weights[self.action_dim + ind] *= w
        discounts = discount ** torch.arange(self.horizon, dtype=torch.float)
        discounts = discounts / discounts.mean()
        loss_weights = discounts[:, None] * dim_weights[None, :]
        loss_weights[0, :self.action_dim] = action_weight
        return loss_weights

    def predict_start_from_noise(self, x_t, t, noise):
        if self.predict_epsilon:
            return (
                extract(self.sqrt_recip_alphas_cumprod, t, x_t.shape) * x_t -
                extract(self.sqrt_recipm1_alphas_cumprod, t, x_t.shape) * noise
            )
        return noise

    def q_posterior(self, x_start, x_t, t):
        posterior_mean = (
            extract(self.posterior_mean_coef1, t, x_t.shape) * x_start +
            extract(self.posterior_mean_coef2, t, x_t.shape) * x_t
        )
        return posterior_mean, extract(self.posterior_variance, t, x_t.shape), \
            extract(self.posterior_log_variance_clipped, t, x_t.shape)

    def p_mean_variance(self, x, cond, t):
        x_recon = self.predict_start_from_noise(x, t=t, noise=self.model(x, cond, t))
        if self.clip_denoised:
            x_recon.clamp_(-1., 1.)
        return self.q_posterior(x_start=x_recon, x_t=x, t=t)

    @torch.no_grad()
    def p_sample_loop(self, shape, cond, return_chain: bool = False, sample_fn=default_sample_fn,
                      plan_size: int = 1, **sample_kwargs):
        device = self.betas.device
        batch_size = shape[0]
        x = torch.randn(shape, device=device)
        x = apply_conditioning(x, cond, self.action_dim)
        chain = [x] if return_chain else None
        values = torch.zeros(batch_size, device=device)
        for i in reversed(range(self.n_timesteps)):
            t = torch.full((batch_size, ), i, device=device, dtype=torch.long)
            x, values = sample_fn(self, x, cond, t, **sample_kwargs)
            x = apply_conditioning(x, cond, self.action_dim)
            if return_chain:
                chain.append(x)
        values = values.reshape(-1, plan_size, *values.shape[1:])
        x = x.reshape(-1, plan_size, *x.shape[1:])
        if plan_size > 1:
            inds = torch.argsort(values, dim=1, descending=True)
            x = x[torch.arange(x.size(0)).unsqueeze(1), inds]
            values = values[torch.arange(values.size(0)).unsqueeze(1), inds]
        if return_chain:
            chain = torch.stack(chain, dim=1)
        return Sample(x, values, chain)

    @torch.no_grad()
    def conditional_sample(self, cond, horizon: Optional[int] = None, **sample_kwargs):
        batch_size = len(cond[0])
        shape = (batch_size, horizon or self.horizon, self.transition_dim)
        return self.p_sample_loop(shape, cond, **sample_kwargs)

    def q_sample(self, x_start, t, noise=None):
        if noise is None:
            noise = torch.randn_like(x_start)
        return (
            extract(self.sqrt_alphas_cumprod, t, x_start.shape) * x_start +
            extract(self.sqrt_one_minus_alphas_cumprod, t, x_start.shape) * noise
        )

    def p_losses(self, x_start, cond, t):
        noise = torch.randn_like(x_start)
        x_noisy = apply_conditioning(self.q_sample(x_start, t, noise), cond, self.action_dim)
        x_recon = apply_conditioning(self.model(x_noisy, cond, t), cond, self.action_dim)
        target = noise if self.predict_epsilon else x_start
        loss = F.mse_loss(x_recon, target, reduction='none')
        lw = self.loss_weights.to(loss.device)
        a0_loss = (loss[:, 0, :self.action_dim] / lw[0, :self.action_dim]).mean()
        loss = (loss * lw).mean()
        return loss, a0_loss

    def forward(self, cond, *args, **kwargs):
        return self.conditional_sample(cond, *args, **kwargs)


class ValueDiffusion(GaussianDiffusion):
    """Value model trained on noised trajectories (guidance critic)."""

    def p_losses(self, x_start, cond, target, t):
        noise = torch.randn_like(x_start)
        x_noisy = apply_conditioning(self.q_sample(x_start, t, noise), cond, self.action_dim)
        pred = self.model(x_noisy, cond, t)
        loss = F.mse_loss(pred, target, reduction='none').mean()
        log = {
            'mean_pred': pred.mean().item(), 'max_pred': pred.max().item(), 'min_pred': pred.min().item(),
        }
        return loss, log

    def forward(self, x, cond, t):
        return self.model(x, cond, t)


@MODEL_REGISTRY.register('pd')
class PlanDiffuser(nn.Module):
    """Diffuser: trajectory diffusion + optional value-guided sampling."""

    def __init__(self, diffuser_model: str, diffuser_model_cfg: dict, value_model: Optional[str],
                 value_model_cfg: Optional[dict], **sample_kwargs):
        super().__init__()
        diffuser_cls = {'GaussianDiffusion': GaussianDiffusion}.get(diffuser_model, GaussianDiffusion) \
            if isinstance(diffuser_model, str) else diffuser_model
        self.diffuser = diffuser_cls(**diffuser_model_cfg)
        self.value = None
        if value_model:
            value_cls = {'ValueDiffusion': ValueDiffusion}.get(value_model, ValueDiffusion) \
                if isinstance(value_model, str) else value_model
            self.value = value_cls(**value_model_cfg)
        self.sample_kwargs = sample_kwargs

    def diffuser_loss(self, x_start, cond, t):
        return self.diffuser.p_losses(x_start, cond, t)

    def value_loss(self, x_start, cond, target, t):
        return self.value.p_losses(x_start, cond, target, t)

    def get_eval(self, cond, batch_size: int = 1):
        cond = self.repeat_cond(cond, batch_size)
        if self.value:
            samples = self.diffuser(
                cond, sample_fn=n_step_guided_p_sample, plan_size=batch_size, guide=self.value,
                **self.sample_kwargs
            )
            actions = samples.trajectories[:, :, :, :self.diffuser.action_dim]
            return actions[:, 0, 0]
        samples = self.diffuser(cond, plan_size=batch_size)
        return samples.trajectories[:, :, :, self.diffuser.action_dim:].squeeze(1)

    def repeat_cond(self, cond, batch_size):
        return {k: v.repeat_interleave(batch_size, dim=0) for k, v in cond.items()}


class ARInvModel(nn.Module):
    """Autoregressive inverse dynamics: discretized per-dim action bins."""

    def __init__(self, hidden_dim: int, obs_dim: int, action_dim: int, low_act: float = -1.0, up_act: float = 1.0):
        super().__init__()
        self.obs_dim = obs_dim
        self.action_dim = action_dim
        self.bins = 80
        self.low_act, self.up_act = low_act, up_act
        self.bin_size = (up_act - low_act) / self.bins
        self.ce = nn.CrossEntropyLoss()
        self.state_embed = nn.Sequential(
            nn.Linear(2 * obs_dim, hidden_dim), nn.ReLU(), nn.Linear(hidden_dim, hidden_dim), nn.ReLU(),
            nn.Linear(hidden_dim, hidden_dim)
        )
        self.lin_mod = nn.ModuleList([nn.Linear(i, hidden_dim) for i in range(1, action_dim)])
        self.act_mod = nn.ModuleList(
            [nn.Sequential(nn.Linear(hidden_dim, hidden_dim), nn.ReLU(), nn.Linear(hidden_dim, self.bins))]
        )
        for _ in range(1, action_dim):
            self.act_mod.append(
                nn.Sequential(nn.Linear(2 * hidden_dim, hidden_dim), nn.ReLU(), nn.Linear(hidden_dim, self.bins))
            )

    def forward(self, comb_state, deterministic: bool = False):
        state_inp = comb_state
        state_d = self.state_embed(state_inp)
        lp_0 = self.act_mod[0](state_d)
        l_0 = torch.distributions.Categorical(logits=lp_0).sample()
        if deterministic:
            a_0 = self.low_act + (l_0 + 0.5) * self.bin_size
        else:
            a_0 = torch.distributions.Uniform(
                self.low_act + l_0 * self.bin_size, self.low_act + (l_0 + 1) * self.bin_size
            ).sample()
        a = [a_0.unsqueeze(1)]
        for i in range(1, self.action_dim):
            lp_i = self.act_mod[i](torch.cat([state_d, self.lin_mod[i - 1](torch.cat(a, dim=1))], dim=1))
            l_i = torch.distributions.Categorical(logits=lp_i).sample()
            if deterministic:
                a_i = self.low_act + (l_i + 0.5) * self.bin_size
            else:
                a_i = torch.distributions.Uniform(
                    self.low_act + l_i * self.bin_size, self.low_act + (l_i + 1) * self.bin_size
                ).sample()
            a.append(a_i.unsqueeze(1))
        return torch.cat(a, dim=1)

    def calc_loss(self, comb_state, action):
        eps = 1e-8
        state_d = self.state_embed(comb_state)
        l_action = torch.div((action - self.low_act).clamp_(eps, self.up_act - self.low_act - eps),
                             self.bin_size, rounding_mode='floor').long()
        loss = self.ce(self.act_mod[0](state_d), l_action[:, 0])
        for i in range(1, self.action_dim):
            loss += self.ce(
                self.act_mod[i](torch.cat([state_d, self.lin_mod[i - 1](action[:, :i])], dim=1)), l_action[:, i]
            )
        return loss / self.action_dim


@MODEL_REGISTRY.register('dd')
class GaussianInvDynDiffusion(nn.Module):
    """Decision Diffuser: diffusion over OBS-ONLY trajectories conditioned on
    returns (classifier-free guidance), actions recovered by an inverse
    dynamics model over consecutive observations."""

    def __init__(
        self,
        model: str,
        model_cfg: dict,
        horizon: int,
        obs_dim: int,
        action_dim: int,
        n_timesteps: int = 1000,
        hidden_dim: int = 256,
        returns_condition: bool = True,
        ar_inv: bool = False,
        train_only_inv: bool = False,
        predict_epsilon: bool = True,
        condition_guidance_w: float = 0.1,
        loss_discount: float = 1.0,
        clip_denoised: bool = True,
    ):
        super().__init__()
        self.horizon = horizon
        self.obs_dim = obs_dim
        self.action_dim = action_dim
        self.transition_dim = obs_dim  # diffusion runs over observations only
        self.model = _resolve(model)(**model_cfg)
        self.returns_condition = returns_condition
        self.condition_guidance_w = condition_guidance_w
        self.ar_inv = ar_inv
        self.train_only_inv = train_only_inv
        self.predict_epsilon = predict_epsilon
        self.clip_denoised = clip_denoised
        self.n_timesteps = int(n_timesteps)
        if ar_inv:
            self.inv_model = ARInvModel(hidden_dim, obs_dim, action_dim)
        else:
            self.inv_model = nn.Sequential(
                nn.Linear(2 * obs_dim, hidden_dim), nn.ReLU(), nn.Linear(hidden_dim, hidden_dim), nn.ReLU(),
                nn.Linear(hidden_dim, action_dim)
            )
        GaussianDiffusion._register_schedule(self, cosine_beta_schedule(n_timesteps))
        discounts = loss_discount ** torch.arange(horizon, dtype=torch.float)
        discounts = discounts / discounts.mean()
        self.loss_weights = discounts[:, None] * torch.ones(obs_dim)[None, :]

    def predict_start_from_noise(self, x_t, t, noise):
        return GaussianDiffusion.predict_start_from_noise(self, x_t, t, noise)

    def q_posterior(self, x_start, x_t, t):
        return GaussianDiffusion.q_posterior(self, x_start, x_t, t)

    def q_sample(self, x_start, t, noise=None):
        return GaussianDiffusion.q_sample(self, x_start, t, noise)

    def p_mean_variance(self, x, cond, t, returns=None):
        if self.returns_condition:
            # classifier-free guidance: mix conditioned/unconditioned eps
            eps_cond = self.model(x, cond, t, returns=returns, use_dropout=False)
            eps_uncond = self.model(x, cond, t, returns=returns, force_dropout=True)
            epsilon = eps_uncond + self.condition_guidance_w * (eps_cond - eps_uncond)
        else:
            epsilon = self.model(x, cond, t)
        x_recon = self.predict_start_from_noise(x, t=t, noise=epsilon)
        if self.clip_denoised:
            x_recon.clamp_(-1., 1.)
        return self.q_posterior(x_start=x_recon, x_t=x, t=t)

    @torch.no_grad()
    def p_sample_loop(self, shape, cond, returns=None, return_chain: bool = False):
        device = self.betas.device
        batch_size = shape[0]
        x = torch.randn(shape, device=device)
        x = apply_conditioning(x, cond, 0)
        chain = [x] if return_chain else None
        for i in reversed(range(self.n_timesteps)):
            t = torch.full((batch_size, ), i, device=device, dtype=torch.long)
            model_mean, _, model_log_variance = self.p_mean_variance(x, cond, t, returns)
            noise = 0.5 * torch.randn_like(x)
            nonzero_mask = (1 - (t == 0).float()).reshape(batch_size, *((1, ) * (len(x.shape) - 1)))
            x = model_mean + nonzero_mask * (0.5 * model_log_variance).exp() * noise
            x = apply_conditioning(x, cond, 0)
            if return_chain:
                chain.append(x)
        if return_chain:
            chain = torch.stack(chain, dim=1)
        return Sample(x, None, chain)

    @torch.no_grad()
    def conditional_sample(self, cond, returns=None, horizon: Optional[int] = None, **kwargs):
        batch_size = len(cond[0])
        shape = (batch_size, horizon or self.horizon, self.obs_dim)
        return self.p_sample_loop(shape, cond, returns, **kwargs)

    def p_losses(self, x_start, cond, t, returns=None):
        noise = torch.randn_like(x_start)
        x_noisy = apply_conditioning(self.q_sample(x_start, t, noise), cond, 0)
        x_recon = self.model(x_noisy, cond, t, returns=returns)
        if not self.predict_epsilon:
            x_recon = apply_conditioning(x_recon, cond, 0)
        target = noise if self.predict_epsilon else x_start
        loss = F.mse_loss(x_recon, target, reduction='none')
        loss = (loss * self.loss_weights.to(loss.device)).mean()
        return loss

    def forward(self, cond, *args, **kwargs):
        return self.conditional_sample(cond, *args, **kwargs)
