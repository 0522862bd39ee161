"""Model wrappers: runtime-composable sampling / state / target-net behavior.

Parity: reference ding/model/wrapper/model_wrappers.py (HiddenStateWrapper:100,
EpsGreedySampleWrapper:561, MultinomialSampleWrapper:526, ActionNoiseWrapper:831,
TargetNetworkWrapper:899, wrapper_name_map:967, model_wrap:993).
"""
import copy
from typing import Any, Callable, Dict, List, Optional, Union

import numpy as np
import torch
import torch.nn as nn

from ding.torch_utils import get_tensor_data, zeros_like
from ding.rl_utils import create_noise_generator


class IModelWrapper(nn.Module):
    """Delegating wrapper base: attribute access falls through to the model."""

    def __init__(self, model: nn.Module):
        super().__init__()
        self._model = model

    def __getattr__(self, key: str) -> Any:
        if key in ['_model'] or key.startswith('_') or key in self.__dict__:
            return super().__getattr__(key)
        try:
            return super().__getattr__(key)
        except AttributeError:
            return getattr(self._model, key)

    @property
    def model(self):
        return self._model

    def state_dict(self, *args, **kwargs):
        # delegate to the innermost raw model so checkpoints are wrapper-free
        return self._model.state_dict(*args, **kwargs)

    def load_state_dict(self, state_dict, *args, **kwargs):
        return self._model.load_state_dict(state_dict, *args, **kwargs)

    def forward(self, *args, **kwargs):
        return self._model.forward(*args, **kwargs)

    def reset(self, data_id: Optional[List[int]] = None, **kwargs):
        if hasattr(self._model, 'reset') and not isinstance(self._model, nn.Module) or isinstance(self._model, IModelWrapper):
            return self._model.reset(data_id=data_id, **kwargs)

    def info(self, attr_name: str = '') -> str:
        inner = self._model.info(attr_name) if isinstance(self._model, IModelWrapper) else type(self._model).__name__
        return f"{type(self).__name__}({inner})"


class BaseModelWrapper(IModelWrapper):
    pass


class HiddenStateWrapper(IModelWrapper):
    """Maintain per-env RNN hidden state across inference calls.

    state_num: number of parallel envs; save_prev_state adds 'prev_state' to
    the output so the transition stores the pre-step state (R2D2).
    """

    def __init__(
        self,
        model: nn.Module,
        state_num: int,
        save_prev_state: bool = False,
        init_fn: Callable = lambda: None,
    ):
        super().__init__(model)
        self._state_num = state_num
        self._state = {i: init_fn() for i in range(state_num)}
        self._save_prev_state = save_prev_state
        self._init_fn = init_fn

    def forward(self, data, **kwargs):
        state_id = kwargs.pop('data_id', None)
        valid_id = state_id if state_id is not None else list(range(self._state_num))
        prev_state = [self._state[i] for i in valid_id]
        if isinstance(data, dict):
            data = {**data, 'prev_state': prev_state}
        else:
            data = {'obs': data, 'prev_state': prev_state}
        output = self._model.forward(data, **kwargs)
        h = output['next_state']
        for i, idx in enumerate(valid_id):
            self._state[idx] = h[i]
        if self._save_prev_state:
            output['prev_state'] = prev_state
        return output

    def reset(self, *args, **kwargs):
        state = kwargs.pop('state', None)
        state_id = kwargs.get('data_id', None)
        self.reset_state(state, state_id)
        if hasattr(self._model, 'reset') and isinstance(self._model, IModelWrapper):
            return self._model.reset(*args, **kwargs)

    def reset_state(self, state: Optional[list] = None, state_id: Optional[List[int]] = None):
        if state_id is None:
            state_id = list(range(self._state_num))
        if state is None:
            state = [self._init_fn() for _ in state_id]
        assert len(state) == len(state_id)
        for idx, s in zip(state_id, state):
            self._state[idx] = s


def sample_action(logit: Optional[torch.Tensor] = None, prob: Optional[torch.Tensor] = None):
    """Multinomial sample supporting trailing action dims."""
    if prob is None:
        prob = torch.softmax(logit, dim=-1)
    shape = prob.shape
    prob = prob.reshape(-1, shape[-1]) + 1e-8
    action = torch.multinomial(prob, 1).squeeze(-1)
    return action.reshape(shape[:-1])


class ArgmaxSampleWrapper(IModelWrapper):

    def forward(self, *args, **kwargs):
        output = self._model.forward(*args, **kwargs)
        logit = output['logit']
        if isinstance(logit, torch.Tensor):
            logit = [logit]
            single = True
        else:
            single = False
        mask = output.get('action_mask')
        if mask is not None and isinstance(mask, torch.Tensor):
            mask = [mask]
        action = []
        for i, l in enumerate(logit):
            if mask is not None:
                l = l.masked_fill(~mask[i].bool(), -1e8) if mask[i].dtype != torch.bool else l.masked_fill(~mask[i], -1e8)
            action.append(l.argmax(dim=-1))
        output['action'] = action[0] if single else action
        return output


class MultinomialSampleWrapper(IModelWrapper):

    def forward(self, *args, **kwargs):
        alpha = kwargs.pop('alpha', None)
        output = self._model.forward(*args, **kwargs)
        logit = output['logit']
        single = isinstance(logit, torch.Tensor)
        logits = [logit] if single else logit
        actions = []
        for l in logits:
            if alpha is not None:
                prob = torch.softmax(l / alpha, dim=-1)
                actions.append(sample_action(prob=prob))
            else:
                actions.append(sample_action(logit=l))
        output['action'] = actions[0] if single else actions
        return output


class EpsGreedySampleWrapper(IModelWrapper):
    """Epsilon-greedy over logits; supports action_mask in the output."""

    def forward(self, *args, **kwargs):
        eps = kwargs.pop('eps')
        output = self._model.forward(*args, **kwargs)
        logit = output['logit']
        single = isinstance(logit, torch.Tensor)
        logits = [logit] if single else logit
        mask = output.get('action_mask')
        if mask is not None and isinstance(mask, torch.Tensor):
            mask = [mask]
        actions = []
        for i, l in enumerate(logits):
            if mask is not None:
                m = mask[i].bool()
                l = l.masked_fill(~m, -1e8)
            greedy = l.argmax(dim=-1)
            if mask is not None:
                prob = mask[i].float() / mask[i].float().sum(-1, keepdim=True).clamp(min=1)
                rand = sample_action(prob=prob)
            else:
                rand = torch.randint_like(greedy, l.shape[-1])
            take_rand = torch.rand(greedy.shape, device=greedy.device) < eps
            actions.append(torch.where(take_rand, rand, greedy))
        output['action'] = actions[0] if single else actions
        return output


class EpsGreedyMultinomialSampleWrapper(IModelWrapper):

    def forward(self, *args, **kwargs):
        eps = kwargs.pop('eps')
        alpha = kwargs.pop('alpha', None)
        output = self._model.forward(*args, **kwargs)
        logit = output['logit']
        single = isinstance(logit, torch.Tensor)
        logits = [logit] if single else logit
        actions = []
        for l in logits:
            if alpha is not None:
                sampled = sample_action(prob=torch.softmax(l / alpha, dim=-1))
            else:
                sampled = sample_action(logit=l)
            rand = torch.randint_like(sampled, l.shape[-1])
            take_rand = torch.rand(sampled.shape, device=sampled.device) < eps
            actions.append(torch.where(take_rand, rand, sampled))
        output['action'] = actions[0] if single else actions
        return output


class DeterministicSampleWrapper(IModelWrapper):
    """action = mu (continuous eval)."""

    def forward(self, *args, **kwargs):
        output = self._model.forward(*args, **kwargs)
        output['action'] = output['logit'][0] if isinstance(output['logit'], list) else output['logit']['mu']
        return output


class ReparamSampleWrapper(IModelWrapper):
    """action ~ N(mu, sigma) (SAC collect)."""

    def forward(self, *args, **kwargs):
        output = self._model.forward(*args, **kwargs)
        logit = output['logit']
        mu, sigma = (logit[0], logit[1]) if isinstance(logit, (list, tuple)) else (logit['mu'], logit['sigma'])
        dist = torch.distributions.Independent(torch.distributions.Normal(mu, sigma), 1)
        output['action'] = dist.sample()
        return output


class HybridArgmaxSampleWrapper(IModelWrapper):

    def forward(self, *args, **kwargs):
        output = self._model.forward(*args, **kwargs)
        logit = output['logit']
        if isinstance(logit, dict):  # VAC-style {'action_type','action_args'}
            action_type = logit['action_type'].argmax(dim=-1)
            action_args = logit['action_args']['mu'] if isinstance(logit['action_args'], dict) \
                else output['action_args']
        else:  # PDQN-style flat Q logits + separate args
            action_type = logit.argmax(dim=-1)
            action_args = output['action_args']
        output['action'] = {'action_type': action_type, 'action_args': action_args}
        return output


class HybridEpsGreedySampleWrapper(IModelWrapper):

    def forward(self, *args, **kwargs):
        eps = kwargs.pop('eps', 0.0)
        output = self._model.forward(*args, **kwargs)
        logit = output['logit']['action_type'] if isinstance(output['logit'], dict) else output['logit']
        greedy = logit.argmax(dim=-1)
        rand = torch.randint_like(greedy, logit.shape[-1])
        take_rand = torch.rand(greedy.shape, device=greedy.device) < eps
        action_type = torch.where(take_rand, rand, greedy)
        action_args = output.get('action_args')
        if action_args is None and isinstance(output['logit'], dict):
            aa = output['logit']['action_args']
            action_args = aa['mu'] if isinstance(aa, dict) else aa
        output['action'] = {'action_type': action_type, 'action_args': action_args}
        return output


class HybridEpsGreedyMultinomialSampleWrapper(HybridEpsGreedySampleWrapper):
    pass


class HybridReparamMultinomialSampleWrapper(IModelWrapper):

    def forward(self, *args, **kwargs):
        output = self._model.forward(*args, **kwargs)
        logit = output['logit']
        action_type = sample_action(logit=logit['action_type'])
        aa = logit['action_args']
        dist = torch.distributions.Normal(aa['mu'], aa['sigma'])
        output['action'] = {'action_type': action_type, 'action_args': dist.sample()}
        return output


class HybridDeterministicArgmaxSampleWrapper(IModelWrapper):

    def forward(self, *args, **kwargs):
        output = self._model.forward(*args, **kwargs)
        logit = output['logit']
        action_type = logit['action_type'].argmax(dim=-1)
        aa = logit['action_args']
        output['action'] = {'action_type': action_type, 'action_args': aa['mu'] if isinstance(aa, dict) else aa}
        return output


class ActionNoiseWrapper(IModelWrapper):
    """Additive exploration noise on continuous actions (DDPG/TD3)."""

    def __init__(
        self,
        model: nn.Module,
        noise_type: str = 'gauss',
        noise_kwargs: dict = {},
        noise_range: Optional[dict] = None,
        action_range: Optional[dict] = {'min': -1, 'max': 1},
    ):
        super().__init__(model)
        self.noise_generator = create_noise_generator(noise_type, noise_kwargs)
        self.noise_range = noise_range
        self.action_range = action_range

    def forward(self, *args, **kwargs):
        # collectors may pass sample-wrapper kwargs irrelevant here
        kwargs.pop('eps', None)
        output = self._model.forward(*args, **kwargs)
        if 'action' in output or 'action_args' in output:
            key = 'action' if 'action' in output else 'action_args'
            action = output[key]
            assert isinstance(action, torch.Tensor)
            action = self.add_noise(action)
            output[key] = action
        return output

    def add_noise(self, action: torch.Tensor) -> torch.Tensor:
        noise = self.noise_generator(action.shape, action.device)
        if self.noise_range is not None:
            noise = noise.clamp(self.noise_range['min'], self.noise_range['max'])
        action = action + noise
        if self.action_range is not None:
            action = action.clamp(self.action_range['min'], self.action_range['max'])
        return action


class TargetNetworkWrapper(IModelWrapper):
    """Target network with 'assign' (periodic copy) or 'momentum' (EMA)
    update. The wrapped model IS the target copy; callers invoke
    ``update(state_dict)`` from the live model."""

    def __init__(self, model: nn.Module, update_type: str, update_kwargs: dict):
        super().__init__(model)
        assert update_type in ('momentum', 'assign')
        self._update_type = update_type
        self._update_kwargs = update_kwargs
        self._update_count = 0

    def reset(self, *args, **kwargs):
        target_update_count = kwargs.pop('target_update_count', None)
        if target_update_count is not None:
            self._update_count = target_update_count
        if isinstance(self._model, IModelWrapper):
            return self._model.reset(*args, **kwargs)

    def update(self, state_dict: dict, direct: bool = False) -> None:
        if direct:
            self._model.load_state_dict(state_dict, strict=True)
            self._update_count = 0
            return
        if self._update_type == 'assign':
            if (self._update_count + 1) % self._update_kwargs['freq'] == 0:
                self._model.load_state_dict(state_dict, strict=True)
            self._update_count += 1
        else:
            theta = self._update_kwargs['theta']
            with torch.no_grad():
                for name, p in self._model.named_parameters():
                    p.data = (1 - theta) * p.data + theta * state_dict[name]


class TeacherNetworkWrapper(IModelWrapper):

    def __init__(self, model, teacher_cfg):
        super().__init__(model)
        self._teacher_cfg = teacher_cfg


class TransformerInputWrapper(IModelWrapper):
    """Maintain a sliding window of the last ``seq_len`` observations per env
    and feed the model sequences (GTrXL collect path)."""

    def __init__(self, model: nn.Module, seq_len: int, init_fn: Callable = lambda: None):
        super().__init__(model)
        self.seq_len = seq_len
        self.obs_memory = None
        self._init_fn = init_fn

    def forward(self, input_obs: torch.Tensor, only_last_logit: bool = True, data_id: Optional[List] = None, **kwargs):
        B = input_obs.shape[0]
        if self.obs_memory is None:
            self.obs_memory = torch.zeros(self.seq_len, B, *input_obs.shape[1:], device=input_obs.device)
            self.memory_idx = [0 for _ in range(B)]
        if data_id is None:
            data_id = list(range(B))
        for i in data_id:
            idx = min(self.memory_idx[i], self.seq_len - 1)
            if self.memory_idx[i] >= self.seq_len:
                self.obs_memory[:, i] = torch.roll(self.obs_memory[:, i], -1, 0)
            self.obs_memory[idx, i] = input_obs[i]
            if self.memory_idx[i] < self.seq_len:
                self.memory_idx[i] += 1
        out = self._model.forward(self.obs_memory, **kwargs)
        if only_last_logit:
            # pick each env's latest step logit
            logits = out['logit']  # [T, B, N]
            out['logit'] = torch.stack([logits[min(self.memory_idx[i], self.seq_len) - 1, i] for i in range(B)])
        out['input_seq'] = self.obs_memory
        return out

    def reset(self, *args, **kwargs):
        state_id = kwargs.get('data_id', None)
        input_seq = kwargs.get('input_seq', None)
        if state_id is None:
            self.obs_memory = None
        else:
            if self.obs_memory is not None:
                for i in state_id:
                    self.obs_memory[:, i] = 0 if input_seq is None else input_seq[:, i]
                    self.memory_idx[i] = 0
        if isinstance(self._model, IModelWrapper):
            return self._model.reset(*args, **kwargs)


class TransformerSegmentWrapper(IModelWrapper):
    """Split long sequences into seq_len segments and concatenate outputs."""

    def __init__(self, model: nn.Module, seq_len: int):
        super().__init__(model)
        self.seq_len = seq_len

    def forward(self, obs: torch.Tensor, **kwargs):
        chunks = list(torch.split(obs, self.seq_len, dim=0))
        outputs = [self._model.forward(c, **kwargs) for c in chunks]
        out = {}
        for k in outputs[0].keys():
            vals = [o[k] for o in outputs if o[k] is not None]
            out[k] = torch.cat(vals, dim=0) if isinstance(vals[0], torch.Tensor) else vals
        return out


class TransformerMemoryWrapper(IModelWrapper):
    """Snapshot/restore GTrXL memory around collect segments."""

    def __init__(self, model: nn.Module, batch_size: int):
        super().__init__(model)
        self._model.reset_memory(batch_size=batch_size)
        self.memory = self._model.get_memory()

    def forward(self, *args, **kwargs):
        out = self._model.forward(*args, **kwargs)
        self.memory = self._model.get_memory()
        return out

    def reset(self, *args, **kwargs):
        state_id = kwargs.get('data_id', None)
        if state_id is None:
            self._model.reset_memory(batch_size=self.memory.shape[2] if self.memory is not None else None)
        else:
            mem = self._model.get_memory()
            if mem is not None:
                mem = mem.clone()
                mem[:, :, state_id] = 0
                self._model.reset_memory(state=mem)
        self.memory = self._model.get_memory()
        if isinstance(self._model, IModelWrapper):
            return self._model.reset(*args, **kwargs)

    def show_memory_occupancy(self, layer: int = 0):
        mem = self.memory
        if mem is None:
            return 0.0
        return (mem[layer].abs().sum(-1) > 0).float().mean().item()


class CombinationArgmaxSampleWrapper(IModelWrapper):
    """PC-BFS style combinatorial action argmax."""

    def forward(self, shot_number, *args, **kwargs):
        output = self._model.forward(shot_number, *args, **kwargs)
        output['action'] = output['logit'].argmax(dim=-1)
        return output


class CombinationMultinomialSampleWrapper(IModelWrapper):

    def forward(self, shot_number, *args, **kwargs):
        output = self._model.forward(shot_number, *args, **kwargs)
        output['action'] = sample_action(logit=output['logit'])
        return output


wrapper_name_map = {
    'base': BaseModelWrapper,
    'hidden_state': HiddenStateWrapper,
    'argmax_sample': ArgmaxSampleWrapper,
    'hybrid_argmax_sample': HybridArgmaxSampleWrapper,
    'eps_greedy_sample': EpsGreedySampleWrapper,
    'eps_greedy_multinomial_sample': EpsGreedyMultinomialSampleWrapper,
    'deterministic_sample': DeterministicSampleWrapper,
    'reparam_sample': ReparamSampleWrapper,
    'hybrid_eps_greedy_sample': HybridEpsGreedySampleWrapper,
    'hybrid_eps_greedy_multinomial_sample': HybridEpsGreedyMultinomialSampleWrapper,
    'hybrid_reparam_multinomial_sample': HybridReparamMultinomialSampleWrapper,
    'hybrid_deterministic_argmax_sample': HybridDeterministicArgmaxSampleWrapper,
    'multinomial_sample': MultinomialSampleWrapper,
    'action_noise': ActionNoiseWrapper,
    'transformer_input': TransformerInputWrapper,
    'transformer_segment': TransformerSegmentWrapper,
    'transformer_memory': TransformerMemoryWrapper,
    'target': TargetNetworkWrapper,
    'teacher': TeacherNetworkWrapper,
    'combination_argmax_sample': CombinationArgmaxSampleWrapper,
    'combination_multinomial_sample': CombinationMultinomialSampleWrapper,
}


def model_wrap(model: Union[nn.Module, IModelWrapper], wrapper_name: str = None, **kwargs) -> IModelWrapper:
    if wrapper_name not in wrapper_name_map:
        raise KeyError(f"unknown model wrapper: {wrapper_name}")
    return wrapper_name_map[wrapper_name](model, **kwargs)


def register_wrapper(name: str, wrapper_type: type) -> None:
    """Register a user wrapper so ``model_wrap(model, name)`` can build it
    (reference model_wrappers.py:1014)."""
    assert isinstance(name, str) and issubclass(wrapper_type, IModelWrapper)
    wrapper_name_map[name] = wrapper_type
