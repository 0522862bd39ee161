"""DreamerV3 policy: actor-critic trained entirely in RSSM latent
imagination; collect/eval run the RSSM filter + actor online.

Parity: reference ding/policy/mbpolicy/dreamer.py ('dreamer':17) and
mbpolicy/utils.py (imagine:43, compute_target:62, compute_actor_loss:80,
RewardEMA:121, tensorstats:139).
"""
from collections import namedtuple
from typing import Any, Dict, List, Union

import torch
import torch.nn as nn

from ding.model import model_wrap
from ding.rl_utils import generalized_lambda_returns, get_train_sample
from ding.torch_utils import to_device
from ding.torch_utils.network.dreamer import static_scan
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate
from .base_policy import Policy


class RewardEMA:
    """EMA of the 5%/95% return quantiles for advantage normalization."""

    def __init__(self, device, alpha: float = 1e-2):
        self.values = torch.zeros(2, device=device)
        self.alpha = alpha
        self.range = torch.tensor([0.05, 0.95], device=device)

    def __call__(self, x: torch.Tensor):
        flat_x = torch.flatten(x.detach())
        x_quantile = torch.quantile(input=flat_x, q=self.range)
        self.values = self.alpha * x_quantile + (1 - self.alpha) * self.values
        scale = torch.clip(self.values[1] - self.values[0], min=1.0)
        return self.values[0].detach(), scale.detach()


def tensorstats(tensor: torch.Tensor, prefix: str) -> Dict[str, float]:
    return {
        f'{prefix}_mean': float(tensor.mean().detach()),
        f'{prefix}_std': float(tensor.std().detach()),
        f'{prefix}_min': float(tensor.min().detach()),
        f'{prefix}_max': float(tensor.max().detach()),
    }


def imagine(cfg, world_model, start: dict, actor, horizon: int):
    """Unroll the actor through RSSM img_steps from posterior `start`
    ({k: [B,T,...]}, flattened to [B*T])."""
    dynamics = world_model.dynamics
    flatten = lambda x: x.reshape([-1] + list(x.shape[2:]))
    start = {k: flatten(v) for k, v in start.items()}

    def step(prev, _):
        state, _, _ = prev
        feat = dynamics.get_feat(state)
        action = actor(feat.detach()).sample()
        succ = dynamics.img_step(state, action, sample=cfg.imag_sample)
        return succ, feat, action

    succ, feats, actions = static_scan(step, [torch.arange(horizon)], (start, None, None))
    states = {k: torch.cat([start[k][None], v[:-1]], 0) for k, v in succ.items()}
    return feats, states, actions


def compute_target(cfg, world_model, critic, imag_feat, imag_state, reward):
    if 'discount' in world_model.heads:
        inp = world_model.dynamics.get_feat(imag_state)
        discount = cfg.discount * world_model.heads['discount'](inp).mean
        discount = discount.detach()
    else:
        discount = cfg.discount * torch.ones_like(reward)
    value = critic(imag_feat).mode()
    target = generalized_lambda_returns(value.squeeze(-1), reward[:-1].squeeze(-1),
                                        discount[:-1].squeeze(-1), cfg.lambda_)
    target = target.unsqueeze(-1)
    weights = torch.cumprod(torch.cat([torch.ones_like(discount[:1]), discount[:-1]], 0), 0).detach()
    return target, weights, value[:-1]


def compute_actor_loss(cfg, actor, reward_ema, imag_feat, imag_action, target, weights, base):
    metrics = {}
    policy = actor(imag_feat.detach())
    actor_ent = policy.entropy()
    if cfg.reward_EMA:
        offset, scale = reward_ema(target)
        normed_target = (target - offset) / scale
        normed_base = (base - offset) / scale
        adv = normed_target - normed_base
        metrics.update(tensorstats(normed_target, 'normed_target'))
        metrics['EMA_005'] = float(reward_ema.values[0].detach())
        metrics['EMA_095'] = float(reward_ema.values[1].detach())
    else:
        adv = target - base
    actor_target = adv
    if cfg.actor_entropy > 0:
        ent_bonus = cfg.actor_entropy * actor_ent[:-1][:, :, None]
        actor_target = actor_target + ent_bonus
        metrics['actor_entropy'] = float(ent_bonus.mean().detach())
    actor_loss = -torch.mean(weights[:-1] * actor_target)
    metrics['actor_ent'] = float(actor_ent.mean().detach())
    return actor_loss, metrics


@POLICY_REGISTRY.register('dreamer')
class DREAMERPolicy(Policy):

    config = dict(
        type='dreamer',
        cuda=False,
        on_policy=False,
        priority=False,
        priority_IS_weight=False,
        random_collect_size=2500,
        transition_with_policy_data=False,
        imag_horizon=15,
        model=dict(),
        learn=dict(
            lambda_=0.95,
            grad_clip=100,
            learning_rate=3e-5,
            batch_size=16,
            batch_length=64,
            imag_sample=True,
            slow_value_target=True,
            slow_target_update=1,
            slow_target_fraction=0.02,
            discount=0.997,
            reward_EMA=True,
            actor_entropy=3e-4,
            value_decay=0.0,
        ),
        collect=dict(unroll_len=1, action_size=None, collect_dyn_sample=True),
        eval=dict(evaluator=dict(eval_freq=5000)),
        other=dict(replay_buffer=dict(type='sequence', replay_buffer_size=100000)),
    )

    def default_model(self) -> tuple:
        return 'dreamervac', ['ding.model.template.vac']

    def _init_learn(self) -> None:
        from copy import deepcopy
        self._lambda = self._cfg.learn.lambda_
        self._grad_clip = self._cfg.learn.grad_clip
        self._critic = self._model.critic
        self._actor = self._model.actor
        if self._cfg.learn.slow_value_target:
            self._slow_value = deepcopy(self._critic)
            self._updates = 0
        self._optimizer_value = torch.optim.Adam(self._critic.parameters(), lr=self._cfg.learn.learning_rate)
        self._optimizer_actor = torch.optim.Adam(self._actor.parameters(), lr=self._cfg.learn.learning_rate)
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._learn_model.reset()
        self._forward_learn_cnt = 0
        if self._cfg.learn.reward_EMA:
            self.reward_ema = RewardEMA(device=self._device)

    def _forward_learn(self, start: dict, world_model=None, envstep: int = 0) -> Dict[str, Any]:
        """`start` is the detached RSSM posterior {logit, stoch, deter} from
        world_model.train (NOT transition dicts)."""
        assert world_model is not None
        log_vars = {}
        self._learn_model.train()
        self._update_slow_target()
        self._actor.requires_grad_(True)
        if self._cuda:
            start = to_device(start, self._device)

        imag_feat, imag_state, imag_action = imagine(
            self._cfg.learn, world_model, start, self._actor, self._cfg.imag_horizon
        )
        reward = world_model.heads['reward'](world_model.dynamics.get_feat(imag_state)).mode()
        target, weights, base = compute_target(
            self._cfg.learn, world_model, self._critic, imag_feat, imag_state, reward
        )
        actor_loss, mets = compute_actor_loss(
            self._cfg.learn, self._actor, getattr(self, 'reward_ema', None), imag_feat, imag_action,
            target, weights, base
        )
        log_vars.update(mets)
        self._actor.requires_grad_(False)

        self._critic.requires_grad_(True)
        value_input = imag_feat
        value = self._critic(value_input[:-1].detach())
        value_loss = -value.log_prob(target.detach())
        if self._cfg.learn.slow_value_target:
            slow_target = self._slow_value(value_input[:-1].detach())
            value_loss = value_loss - value.log_prob(slow_target.mode().detach())
        if self._cfg.learn.value_decay:
            value_loss = value_loss + self._cfg.learn.value_decay * value.mode()
        value_loss = torch.mean(weights[:-1] * value_loss[:, :, None])
        self._critic.requires_grad_(False)

        log_vars.update(tensorstats(value.mode(), 'value'))
        log_vars.update(tensorstats(target, 'target'))
        log_vars.update(tensorstats(reward, 'imag_reward'))

        self._model.requires_grad_(True)
        world_model.requires_grad_(True)
        self._optimizer_actor.zero_grad()
        actor_loss.backward()
        actor_norm = nn.utils.clip_grad_norm_(self._actor.parameters(), self._grad_clip)
        self._optimizer_actor.step()
        self._optimizer_value.zero_grad()
        value_loss.backward()
        critic_norm = nn.utils.clip_grad_norm_(self._critic.parameters(), self._grad_clip)
        self._optimizer_value.step()
        self._model.requires_grad_(False)
        world_model.requires_grad_(False)

        self._forward_learn_cnt += 1
        return {
            **log_vars,
            'actor_loss': float(actor_loss.detach()),
            'critic_loss': float(value_loss.detach()),
            'actor_grad_norm': float(actor_norm),
            'critic_grad_norm': float(critic_norm),
        }

    def _update_slow_target(self) -> None:
        if self._cfg.learn.slow_value_target:
            if self._updates % self._cfg.learn.slow_target_update == 0:
                mix = self._cfg.learn.slow_target_fraction
                for s, d in zip(self._critic.parameters(), self._slow_value.parameters()):
                    d.data = mix * s.data + (1 - mix) * d.data
            self._updates += 1

    def _state_dict_learn(self) -> Dict[str, Any]:
        return {
            'model': self._learn_model.state_dict(),
            'optimizer_value': self._optimizer_value.state_dict(),
            'optimizer_actor': self._optimizer_actor.state_dict(),
        }

    def _load_state_dict_learn(self, state_dict: Dict[str, Any]) -> None:
        self._learn_model.load_state_dict(state_dict['model'])
        self._optimizer_value.load_state_dict(state_dict['optimizer_value'])
        self._optimizer_actor.load_state_dict(state_dict['optimizer_actor'])

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._collect_model = model_wrap(self._model, wrapper_name='base')
        self._collect_model.reset()

    def _rssm_act(self, data, world_model, reset, state, sample: bool):
        """Shared collect/eval path: RSSM filter step + actor head."""
        data_id = list(data.keys())
        obs = default_collate(list(data.values()))
        if self._cuda:
            obs = to_device(obs, self._device)
        if state is None:
            B = len(data_id)
            latent = world_model.dynamics.initial(B, obs.device if isinstance(obs, torch.Tensor) else self._device)
            action_size = self._cfg.collect.action_size or world_model.action_size
            action = torch.zeros(B, action_size, device=latent['deter'].device)
        else:
            latent = default_collate([s[0] for s in state])
            action = default_collate([s[1] for s in state])
            if self._cuda:
                latent, action = to_device(latent, self._device), to_device(action, self._device)
            if action.dim() == 1:
                action = action.unsqueeze(-1)
            if reset is not None and reset.any():
                mask = 1 - torch.as_tensor(reset, dtype=action.dtype, device=action.device)
                for k in latent:
                    latent[k] = latent[k] * mask.reshape(-1, *([1] * (latent[k].dim() - 1)))
                action = action * mask.reshape(-1, 1)
        if world_model.obs_type == 'RGB':
            obs = obs - 0.5
        embed = world_model.encoder(obs.float())
        latent, _ = world_model.dynamics.obs_step(latent, action, embed, self._cfg.collect.collect_dyn_sample)
        feat = world_model.dynamics.get_feat(latent)
        dist = self._actor(feat)
        action = dist.sample() if sample else dist.mode()
        logprob = dist.log_prob(action)
        latent = {k: v.detach() for k, v in latent.items()}
        action = action.detach()
        state_out = [( {k: v[i] for k, v in latent.items()}, action[i]) for i in range(len(data_id))]
        act_env = torch.where(action == 1)[1] if world_model.action_type == 'discrete' else action
        output = {'action': act_env, 'logprob': logprob}
        if self._cuda:
            output = to_device(output, 'cpu')
        out = default_decollate(output)
        for i, o in enumerate(out):
            o['state'] = state_out[i]
        return {i: d for i, d in zip(data_id, out)}

    def _forward_collect(self, data: dict, world_model=None, envstep: int = 0, reset=None, state=None, **kwargs):
        self._collect_model.eval()
        with torch.no_grad():
            return self._rssm_act(data, world_model, reset, state, sample=True)

    def _process_transition(self, obs: Any, model_output: dict, timestep: namedtuple) -> dict:
        return {
            'obs': obs,
            'action': model_output['action'],
            'reward': timestep.reward,
            'discount': 1. - timestep.done,
            'done': timestep.done,
        }

    def _get_train_sample(self, data: list) -> Union[None, List[Any]]:
        return get_train_sample(data, self._unroll_len)

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(self._model, wrapper_name='base')
        self._eval_model.reset()

    def _forward_eval(self, data: dict, world_model=None, reset=None, state=None, **kwargs):
        self._eval_model.eval()
        with torch.no_grad():
            return self._rssm_act(data, world_model, reset, state, sample=False)

    def _monitor_vars_learn(self) -> List[str]:
        return [
            'normed_target_mean', 'normed_target_std', 'normed_target_min', 'normed_target_max', 'EMA_005',
            'EMA_095', 'actor_entropy', 'value_mean', 'value_std', 'value_min', 'value_max', 'target_mean',
            'target_std', 'target_min', 'target_max', 'imag_reward_mean', 'imag_reward_std', 'imag_reward_min',
            'imag_reward_max', 'actor_ent', 'actor_loss', 'critic_loss', 'actor_grad_norm', 'critic_grad_norm'
        ]
