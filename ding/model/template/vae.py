"""Action VAE (TD3-VAE / BCQ generative action model).

Parity: reference ding/model/template/vae.py ('vae').
"""
from typing import Dict

import torch
import torch.nn as nn

from ding.utils import MODEL_REGISTRY, squeeze


@MODEL_REGISTRY.register('vae')
class VanillaVAE(nn.Module):
    """Encode (obs, action) -> z; decode (obs, z) -> reconstructed action +
    predicted residual obs delta."""

    def __init__(self, action_shape: int, obs_shape: int, latent_size: int, hidden_size_list=(256, 256), **kwargs):
        super().__init__()
        action_shape, obs_shape = squeeze(action_shape), squeeze(obs_shape)
        self.latent_size = latent_size
        self.action_shape = action_shape
        h = hidden_size_list[0]
        self.encoder = nn.Sequential(nn.Linear(obs_shape + action_shape, h), nn.ReLU(), nn.Linear(h, h), nn.ReLU())
        self.mu_head = nn.Linear(h, latent_size)
        self.logvar_head = nn.Linear(h, latent_size)
        self.decoder = nn.Sequential(nn.Linear(obs_shape + latent_size, h), nn.ReLU(), nn.Linear(h, h), nn.ReLU())
        self.action_head = nn.Sequential(nn.Linear(h, action_shape), nn.Tanh())
        self.residual_head = nn.Linear(h, obs_shape)

    def encode(self, inputs: Dict) -> Dict:
        x = torch.cat([inputs['obs'], inputs['action']], dim=-1)
        e = self.encoder(x)
        return {'mu': self.mu_head(e), 'log_var': self.logvar_head(e)}

    def reparameterize(self, mu: torch.Tensor, log_var: torch.Tensor) -> torch.Tensor:
        std = torch.exp(0.5 * log_var)
        return mu + std * torch.randn_like(std)

    def decode(self, inputs: Dict) -> Dict:
        x = torch.cat([inputs['obs'], inputs['z']], dim=-1)
        d = self.decoder(x)
        return {'reconstruction_action': self.action_head(d), 'predition_residual': self.residual_head(d)}

    def decode_with_obs(self, inputs: Dict) -> Dict:
        return self.decode(inputs)

    def forward(self, inputs: Dict) -> Dict:
        enc = self.encode(inputs)
        z = self.reparameterize(enc['mu'], enc['log_var'])
        dec = self.decode({'obs': inputs['obs'], 'z': z})
        return {
            'recons_action': dec['reconstruction_action'],
            'prediction_residual': dec['predition_residual'],
            'input': inputs['action'],
            'mu': enc['mu'],
            'log_var': enc['log_var'],
            'z': z,
        }

    def loss_function(self, args: Dict, **kwargs) -> Dict:
        recons, inp = args['recons_action'], args['input']
        mu, log_var = args['mu'], args['log_var']
        kld_weight = kwargs.get('kld_weight', 0.01)
        predict_weight = kwargs.get('predict_weight', 0.01)
        recons_loss = torch.nn.functional.mse_loss(recons, inp)
        kld_loss = torch.mean(-0.5 * torch.sum(1 + log_var - mu ** 2 - log_var.exp(), dim=1))
        if 'original_action' in kwargs and 'true_residual' in kwargs:
            predict_loss = torch.nn.functional.mse_loss(args['prediction_residual'], kwargs['true_residual'])
        else:
            predict_loss = torch.zeros(())
        loss = recons_loss + kld_weight * kld_loss + predict_weight * predict_loss
        return {'loss': loss, 'reconstruction_loss': recons_loss, 'kld_loss': kld_loss, 'predict_loss': predict_loss}
