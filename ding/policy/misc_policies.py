"""IBC (implicit BC), BCQ, TD3-VAE, prompt PG / prompt AWR (LLM candidate
policies), procedure cloning.

Parity: reference ding/policy/{ibc,bcq,td3_vae,prompt_pg,prompt_awr,pc}.py.
"""
import copy
from collections import namedtuple
from typing import Any, Dict, List

import torch
import torch.nn.functional as F

from ding.model import model_wrap
from ding.torch_utils import Adam, to_device
from ding.utils import POLICY_REGISTRY
from ding.utils.data import default_collate, default_decollate
from .base_policy import Policy
from .common_utils import default_preprocess_learn
from .ddpg import TD3Policy
from .offline import BehaviourCloningPolicy


@POLICY_REGISTRY.register('ibc')
class IBCPolicy(BehaviourCloningPolicy):
    """Implicit BC: EBM trained with InfoNCE against sampled negatives;
    inference by derivative-free or Langevin optimization."""

    config = dict(
        type='ibc',
        continuous=True,
        model=dict(),
        learn=dict(batch_size=64, learning_rate=1e-3, update_per_collect=1, optim=dict(type='dfo')),
        collect=dict(unroll_len=1),
        eval=dict(),
        other=dict(replay_buffer=dict(replay_buffer_size=10000)),
    )

    def default_model(self) -> tuple:
        return 'ebm', ['ding.model.template.ebm']

    def _init_learn(self) -> None:
        from ding.model.template.ebm import DFO, LangevinMCMC
        self._optimizer = Adam(self._model.parameters(), lr=self._cfg.learn.learning_rate)
        opt_type = self._cfg.learn.optim.get('type', 'dfo')
        self._stochastic_optim = DFO() if opt_type == 'dfo' else LangevinMCMC()
        self._action_dim = self._cfg.model.action_shape
        self._learn_model = self._model
        self._learn_model.train()

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(data)
        if self._cuda:
            data = to_device(data, self._device)
        obs, action = data['obs'], data['action']
        if action.dim() == 1:
            action = action.unsqueeze(-1)
        B = obs.shape[0]
        negatives = self._stochastic_optim.sample(obs, self._model, action.shape[-1])  # [B, n, A]
        candidates = torch.cat([action.unsqueeze(1), negatives], dim=1)  # [B, 1+n, A]
        obs_tiled = obs.unsqueeze(1).expand(B, candidates.shape[1], obs.shape[-1])
        energy = self._model(obs_tiled, candidates)  # [B, 1+n]
        # InfoNCE: ground-truth action (slot 0) should have the LOWEST energy
        logits = -energy
        labels = torch.zeros(B, dtype=torch.long, device=obs.device)
        loss = F.cross_entropy(logits, labels)
        self._optimizer.zero_grad()
        loss.backward()
        if self._cfg.multi_gpu:
            self.sync_gradients(self._model)
        self._optimizer.step()
        return {'cur_lr': self._optimizer.defaults['lr'], 'total_loss': loss.item()}

    def _init_eval(self) -> None:
        self._init_learn_shared_eval()

    def _init_learn_shared_eval(self):
        if not hasattr(self, '_stochastic_optim'):
            from ding.model.template.ebm import DFO
            self._stochastic_optim = DFO()
        self._eval_model = self._model

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._init_learn_shared_eval()
        self._collect_model = self._model

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._model.eval()
        action = self._stochastic_optim.infer(collated, self._model, self._cfg.model.action_shape)
        if self._cuda:
            action = action.cpu()
        out = default_decollate({'action': action})
        return {i: d for i, d in zip(data_id, out)}

    def _forward_collect(self, data: Dict[int, Any], **kwargs) -> Dict[int, Any]:
        return self._forward_eval(data)


@POLICY_REGISTRY.register('bcq')
class BCQPolicy(Policy):
    """Batch-constrained Q: VAE generative actions + perturbation net +
    twin-critic clipped target. Offline only."""

    config = dict(
        type='bcq',
        cuda=False,
        on_policy=False,
        model=dict(),
        learn=dict(
            batch_size=100,
            learning_rate_q=3e-4,
            learning_rate_policy=3e-4,
            learning_rate_vae=3e-4,
            lmbda=0.75,
            phi=0.05,
            target_theta=0.005,
            discount_factor=0.99,
            update_per_collect=1,
        ),
        collect=dict(unroll_len=1, ),
        eval=dict(),
        other=dict(replay_buffer=dict(replay_buffer_size=10000)),
    )

    def default_model(self) -> tuple:
        return 'continuous_qac', ['ding.model.template.qac']

    def _init_learn(self) -> None:
        from ding.model.template.vae import VanillaVAE
        obs_shape = self._cfg.model.obs_shape
        action_shape = self._cfg.model.action_shape
        self._vae = VanillaVAE(action_shape, obs_shape, latent_size=action_shape * 2)
        if self._cuda:
            self._vae.cuda()
        self._optimizer_q = Adam(self._model.critic.parameters(), lr=self._cfg.learn.learning_rate_q)
        self._optimizer_policy = Adam(self._model.actor.parameters(), lr=self._cfg.learn.learning_rate_policy)
        self._optimizer_vae = Adam(self._vae.parameters(), lr=self._cfg.learn.learning_rate_vae)
        self._gamma = self._cfg.learn.discount_factor
        self._lmbda = self._cfg.learn.lmbda
        self._phi = self._cfg.learn.phi
        self._target_model = model_wrap(
            copy.deepcopy(self._model), wrapper_name='target', update_type='momentum',
            update_kwargs={'theta': self._cfg.learn.target_theta}
        )
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._learn_model.train()
        self._target_model.train()

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        data = default_preprocess_learn(data, use_nstep=False)
        if self._cuda:
            data = to_device(data, self._device)
        obs, action = data['obs'], data['action']
        if action.dim() == 1:
            action = action.unsqueeze(-1)
        reward = data['reward'].reshape(-1)
        done = data['done']
        # 1. VAE reconstruction
        vae_out = self._vae({'obs': obs, 'action': action})
        vae_losses = self._vae.loss_function(vae_out)
        self._optimizer_vae.zero_grad()
        vae_losses['loss'].backward()
        self._optimizer_vae.step()
        # 2. critic: sample candidate actions from VAE at next state
        with torch.no_grad():
            B = obs.shape[0]
            rep = 10
            next_obs_rep = data['next_obs'].repeat_interleave(rep, dim=0)
            z = torch.randn(B * rep, self._vae.latent_size, device=obs.device).clamp(-0.5, 0.5)
            gen_action = self._vae.decode({'obs': next_obs_rep, 'z': z})['reconstruction_action']
            target_q = self._target_model.forward(
                {'obs': next_obs_rep, 'action': gen_action}, mode='compute_critic'
            )['q_value']
            q1, q2 = target_q[0], target_q[1]
            target = self._lmbda * torch.min(q1, q2) + (1 - self._lmbda) * torch.max(q1, q2)
            target = target.reshape(B, rep).max(dim=1)[0]
            target = reward + self._gamma * (1 - done) * target
        q_value = self._learn_model.forward({'obs': obs, 'action': action}, mode='compute_critic')['q_value']
        critic_loss = F.mse_loss(q_value[0], target) + F.mse_loss(q_value[1], target)
        self._optimizer_q.zero_grad()
        critic_loss.backward()
        self._optimizer_q.step()
        # 3. actor perturbation: maximize Q on perturbed VAE actions
        with torch.no_grad():
            z = torch.randn(obs.shape[0], self._vae.latent_size, device=obs.device).clamp(-0.5, 0.5)
            sampled = self._vae.decode({'obs': obs, 'z': z})['reconstruction_action']
        perturb = self._learn_model.forward(obs, mode='compute_actor')['action'] * self._phi
        perturbed = (sampled + perturb).clamp(-1, 1)
        q = self._learn_model.forward({'obs': obs, 'action': perturbed}, mode='compute_critic')['q_value'][0]
        actor_loss = -q.mean()
        self._optimizer_policy.zero_grad()
        actor_loss.backward()
        self._optimizer_policy.step()
        self._target_model.update(self._learn_model.state_dict())
        return {
            'total_loss': critic_loss.item() + actor_loss.item(),
            'critic_loss': critic_loss.item(),
            'actor_loss': actor_loss.item(),
            'vae_loss': vae_losses['loss'].item(),
            'cur_lr': self._optimizer_q.defaults['lr'],
        }

    def _init_collect(self) -> None:
        pass

    def _forward_collect(self, data, **kwargs):
        raise NotImplementedError("BCQ is offline-only")

    def _process_transition(self, obs, policy_output, timestep):
        return {'obs': obs, 'next_obs': timestep.obs, 'action': policy_output['action'],
                'reward': timestep.reward, 'done': timestep.done}

    def _get_train_sample(self, transitions):
        return transitions

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(self._model, wrapper_name='base')

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        data_id = list(data.keys())
        collated = default_collate(list(data.values()))
        if self._cuda:
            collated = to_device(collated, self._device)
        self._eval_model.eval()
        with torch.no_grad():
            z = torch.randn(collated.shape[0], self._vae.latent_size, device=collated.device).clamp(-0.5, 0.5)
            sampled = self._vae.decode({'obs': collated, 'z': z})['reconstruction_action']
            perturb = self._eval_model.forward(collated, mode='compute_actor')['action'] * self._phi
            action = (sampled + perturb).clamp(-1, 1)
        if self._cuda:
            action = action.cpu()
        out = default_decollate({'action': action})
        return {i: d for i, d in zip(data_id, out)}


@POLICY_REGISTRY.register('td3_vae')
class TD3VAEPolicy(TD3Policy):
    """TD3 in a VAE action-latent space (HyAR-style)."""

    config = dict(
        type='td3_vae',
        original_action_shape=1,
        model=dict(twin_critic=True, action_space='regression'),
        learn=dict(
            update_per_collect=1, batch_size=128, learning_rate_actor=3e-4, learning_rate_critic=3e-4,
            learning_rate_vae=3e-4, target_theta=0.005, discount_factor=0.99, actor_update_freq=2, noise=True,
            noise_sigma=0.2, noise_range=dict(min=-0.5, max=0.5), warm_up_update=100, ignore_done=False,
        ),
    )

    def _init_learn(self) -> None:
        super()._init_learn()
        from ding.model.template.vae import VanillaVAE
        obs_shape = self._cfg.model.obs_shape
        act = self._cfg.original_action_shape
        latent = self._cfg.model.action_shape  # latent action space = TD3 action space
        self._vae = VanillaVAE(act, obs_shape, latent_size=latent)
        if self._cuda:
            self._vae.cuda()
        self._optimizer_vae = Adam(self._vae.parameters(), lr=self._cfg.learn.learning_rate_vae)

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        """Phase-aware learn (serial_pipeline_td3_vae): warm_up / vae_phase
        batches train the VAE; rl_phase batches run TD3 in the latent action
        space (real actions encoded through the frozen VAE)."""
        if data and isinstance(data[0], dict) and (data[0].get('warm_up') or data[0].get('vae_phase')):
            return self.train_vae(data)
        if data and isinstance(data[0], dict) and data[0].get('rl_phase'):
            data = [dict(d) for d in data]
            need_encode = [d for d in data if 'latent_action' not in d]
            if need_encode:
                with torch.no_grad():
                    obs = torch.stack([torch.as_tensor(d['obs'], dtype=torch.float32) for d in need_encode])
                    act = torch.stack(
                        [torch.as_tensor(d['action'], dtype=torch.float32).reshape(-1) for d in need_encode]
                    )
                    if self._cuda:
                        obs, act = obs.cuda(), act.cuda()
                    latent = self._vae.encode({'obs': obs, 'action': act})['mu'].cpu()
                for d, z in zip(need_encode, latent):
                    d['latent_action'] = z
            for d in data:
                d['action'] = torch.as_tensor(d['latent_action'], dtype=torch.float32).reshape(-1)
                for k in ('warm_up', 'rl_phase', 'vae_phase', 'latent_action'):
                    d.pop(k, None)
        return super()._forward_learn(data)

    def train_vae(self, data: List[Dict[str, Any]]) -> Dict[str, float]:
        collated = default_preprocess_learn(data, use_nstep=False)
        if self._cuda:
            collated = to_device(collated, self._device)
        action = collated['action']
        if action.dim() == 1:
            action = action.unsqueeze(-1)
        out = self._vae({'obs': collated['obs'], 'action': action})
        losses = self._vae.loss_function(out)
        self._optimizer_vae.zero_grad()
        losses['loss'].backward()
        self._optimizer_vae.step()
        return {'vae_loss': losses['loss'].item()}

    def _forward_collect(self, data: Dict[int, Any], **kwargs) -> Dict[int, Any]:
        """TD3 acts in the latent space; the VAE decoder maps the latent to
        the env's real action. Both are kept in the output so learn gets
        the latent and the env gets the real action (HyAR)."""
        out = super()._forward_collect(data, **kwargs)
        with torch.no_grad():
            for env_id, o in out.items():
                z = torch.as_tensor(o['action'], dtype=torch.float32).reshape(1, -1)
                obs = torch.as_tensor(data[env_id], dtype=torch.float32).reshape(1, -1)
                dec = self._vae.decode({'obs': obs, 'z': z})['reconstruction_action']
                o['latent_action'] = o['action']
                o['action'] = dec.reshape(-1).cpu()
        return out

    def _process_transition(self, obs, policy_output, timestep):
        tr = super()._process_transition(obs, policy_output, timestep)
        if 'latent_action' in policy_output:
            tr['latent_action'] = policy_output['latent_action']
        return tr

    def _init_eval(self) -> None:
        super()._init_eval()

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        out = super()._forward_eval(data)
        with torch.no_grad():
            for env_id, o in out.items():
                z = torch.as_tensor(o['action'], dtype=torch.float32).reshape(1, -1)
                obs = torch.as_tensor(data[env_id], dtype=torch.float32).reshape(1, -1)
                o['action'] = self._vae.decode({'obs': obs, 'z': z})['reconstruction_action'].reshape(-1).cpu()
        return out


# -------------------------------------------------------------- LLM prompts
@POLICY_REGISTRY.register('prompt_pg')
class PromptPGPolicy(Policy):
    """Policy gradient over discrete candidate prompts (TabMWP-style):
    the model scores each candidate shot; REINFORCE on episode reward."""

    config = dict(
        type='prompt_pg',
        cuda=False,
        on_policy=True,
        shot_number=1,
        model=dict(),
        learn=dict(batch_size=16, learning_rate=1e-4, entropy_weight=0.001, grad_norm=0.5, ignore_done=False),
        collect=dict(unroll_len=1, discount_factor=1.0, collector=dict(get_train_sample=True, type='episode')),
        eval=dict(),
    )

    def default_model(self) -> tuple:
        return 'language_transformer', ['ding.model.template.language_transformer']

    def _init_learn(self) -> None:
        self._optimizer = Adam(self._model.parameters(), lr=self._cfg.learn.learning_rate,
                               grad_clip_type='clip_norm', clip_value=self._cfg.learn.grad_norm)
        self._entropy_weight = self._cfg.learn.entropy_weight
        self._learn_model = self._model

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        self._model.train()
        total_policy_loss, total_entropy = [], []
        for sample in data:
            output = self._model.forward(sample['obs'], mode='compute_actor')
            logit = output['logit']
            dist = torch.distributions.Categorical(logits=logit)
            ret = torch.as_tensor(sample['return'], dtype=torch.float32)
            logp = dist.log_prob(torch.as_tensor(sample['action']).reshape(-1))
            total_policy_loss.append(-(logp * ret).mean())
            total_entropy.append(dist.entropy().mean())
        policy_loss = torch.stack(total_policy_loss).mean()
        entropy_loss = torch.stack(total_entropy).mean()
        total = policy_loss - self._entropy_weight * entropy_loss
        self._optimizer.zero_grad()
        total.backward()
        self._optimizer.step()
        return {'cur_lr': self._optimizer.defaults['lr'], 'total_loss': total.item(),
                'policy_loss': policy_loss.item(), 'entropy_loss': entropy_loss.item()}

    def _init_collect(self) -> None:
        self._unroll_len = self._cfg.collect.unroll_len
        self._gamma = self._cfg.collect.discount_factor
        self._collect_model = self._model

    def _forward_collect(self, data: Dict[int, Any], **kwargs) -> Dict[int, Any]:
        out = {}
        self._model.eval()
        with torch.no_grad():
            for env_id, obs in data.items():
                o = self._model.forward(obs, mode='compute_actor')
                dist = torch.distributions.Categorical(logits=o['logit'])
                action = dist.sample((self._cfg.shot_number, )).reshape(-1)
                out[env_id] = {'logit': o['logit'], 'action': action}
        return out

    def _process_transition(self, obs, policy_output, timestep) -> Dict[str, Any]:
        return {'obs': obs, 'action': policy_output['action'], 'reward': timestep.reward, 'done': timestep.done}

    def _get_train_sample(self, transitions: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
        R = 0.0
        for t in reversed(transitions):
            R = self._gamma * R + float(torch.as_tensor(t['reward']).reshape(-1)[0])
            t['return'] = R
        return transitions

    def _init_eval(self) -> None:
        self._eval_model = self._model

    def _forward_eval(self, data: Dict[int, Any]) -> Dict[int, Any]:
        out = {}
        self._model.eval()
        with torch.no_grad():
            for env_id, obs in data.items():
                o = self._model.forward(obs, mode='compute_actor')
                k = min(self._cfg.shot_number, o['logit'].shape[-1])
                action = o['logit'].topk(k, dim=-1).indices.reshape(-1)
                out[env_id] = {'logit': o['logit'], 'action': action}
        return out


@POLICY_REGISTRY.register('prompt_awr')
class PromptAWRPolicy(PromptPGPolicy):
    """Advantage-weighted regression over prompt candidates."""

    config = dict(
        type='prompt_awr',
        learn=dict(batch_size=16, learning_rate=1e-4, beta=1.0, weight_max=20.0, entropy_weight=0.001,
                   grad_norm=0.5, value_weight=0.5, ignore_done=False),
    )

    def _forward_learn(self, data: List[Dict[str, Any]]) -> Dict[str, Any]:
        self._model.train()
        losses = []
        for sample in data:
            output = self._model.forward(sample['obs'], mode='compute_actor')
            logit = output['logit']
            dist = torch.distributions.Categorical(logits=logit)
            ret = float(sample['return'])
            adv = ret  # no baseline in the candidate-scoring setting
            weight = min(self._cfg.learn.weight_max, torch.exp(torch.tensor(adv / self._cfg.learn.beta)).item())
            logp = dist.log_prob(torch.as_tensor(sample['action']).reshape(-1))
            losses.append(-(weight * logp).mean())
        loss = torch.stack(losses).mean()
        self._optimizer.zero_grad()
        loss.backward()
        self._optimizer.step()
        return {'cur_lr': self._optimizer.defaults['lr'], 'total_loss': loss.item()}


@POLICY_REGISTRY.register('pc_bfs')
class ProcedureCloningBFSPolicy(BehaviourCloningPolicy):
    """Procedure cloning with BFS intermediate supervision (maze planning):
    supervised on the value-iteration rollout sequence
    (ding/utils/misc_helpers.py get_vi_sequence)."""

    config = dict(
        type='pc_bfs',
        continuous=False,
        learn=dict(batch_size=32, learning_rate=1e-3, update_per_collect=1, weight_decay=1e-4,
                   ce_label_smooth=False, show_accuracy=False, tanh_mask=False, lr_decay=False, momentum=0.9),
    )


@POLICY_REGISTRY.register('IL')
class ILPolicy(Policy):
    """Imitation learning by logit regression: supervise the learner's action
    logits against recorded expert logits (MSE).

    Parity: reference ding/policy/il.py ('IL':15) — generalised: the
    reference hard-codes a football expert; here any expert data with
    'obs'/'logit' fields trains any logit-producing model.
    """

    config = dict(
        type='IL',
        cuda=False,
        on_policy=False,
        priority=False,
        model=dict(),
        learn=dict(multi_gpu=False, update_per_collect=20, batch_size=64, learning_rate=0.0002),
        collect=dict(unroll_len=1, discount_factor=0.99),
        eval=dict(evaluator=dict(eval_freq=800)),
        other=dict(replay_buffer=dict(replay_buffer_size=100000)),
    )

    def default_model(self) -> tuple:
        return 'dqn', ['ding.model.template.q_learning']

    def _init_learn(self) -> None:
        self._optimizer = Adam(self._model.parameters(), lr=self._cfg.learn.learning_rate)
        self._learn_model = model_wrap(self._model, wrapper_name='base')
        self._learn_model.train()
        self._learn_model.reset()
        self._forward_learn_cnt = 0

    def _forward_learn(self, data) -> Dict[str, Any]:
        if isinstance(data, list):
            data = default_collate(data, cat_1dim=False)
        if self._cuda:
            data = to_device(data, self._device)
        obs = data['obs']
        if isinstance(obs, dict):
            obs = obs.get('processed_obs', next(iter(obs.values())))
        logit = data['logit']
        model_logit = self._learn_model.forward(obs.float())['logit']
        supervised_loss = torch.nn.functional.mse_loss(model_logit, logit)
        self._optimizer.zero_grad()
        supervised_loss.backward()
        self._optimizer.step()
        self._forward_learn_cnt += 1
        return {'cur_lr': self._optimizer.defaults['lr'], 'supervised_loss': supervised_loss.item()}

    def _monitor_vars_learn(self):
        return ['cur_lr', 'supervised_loss']

    def _state_dict_learn(self) -> Dict[str, Any]:
        return {'model': self._learn_model.state_dict(), 'optimizer': self._optimizer.state_dict()}

    def _load_state_dict_learn(self, state_dict: Dict[str, Any]) -> None:
        self._learn_model.load_state_dict(state_dict['model'])
        self._optimizer.load_state_dict(state_dict['optimizer'])

    def _init_collect(self) -> None:
        self._collect_model = model_wrap(self._model, wrapper_name='argmax_sample')
        self._collect_model.eval()
        self._collect_model.reset()
        self._gamma = self._cfg.collect.discount_factor
        self._unroll_len = self._cfg.collect.unroll_len

    def _forward_collect(self, data: dict, **kwargs) -> dict:
        data_id = list(data.keys())
        obs = default_collate(list(data.values()))
        if self._cuda:
            obs = to_device(obs, self._device)
        with torch.no_grad():
            output = self._collect_model.forward(obs.float())
        if self._cuda:
            output = to_device(output, 'cpu')
        output = default_decollate(output)
        return {i: d for i, d in zip(data_id, output)}

    def _process_transition(self, obs, model_output, timestep) -> Dict[str, Any]:
        return {
            'obs': obs, 'next_obs': timestep.obs, 'logit': model_output['logit'],
            'action': model_output['action'], 'reward': timestep.reward, 'done': timestep.done,
        }

    def _get_train_sample(self, data):
        from ding.rl_utils import get_train_sample
        return get_train_sample(data, self._cfg.collect.unroll_len)

    def _init_eval(self) -> None:
        self._eval_model = model_wrap(self._model, wrapper_name='argmax_sample')
        self._eval_model.eval()
        self._eval_model.reset()

    def _forward_eval(self, data: dict) -> dict:
        return self._forward_collect(data)


# reference name alias: 'td3-vae' (dash) == 'td3_vae'
from ding.utils import POLICY_REGISTRY as _PR
if 'td3-vae' not in _PR:
    _PR.register('td3-vae')(TD3VAEPolicy)
