"""Trajectory-diffusion building blocks: 1-D temporal UNet denoiser and value
network, cosine beta schedule, conditioning helpers.

Parity: reference ding/torch_utils/network/diffusion.py (extract:11,
cosine_beta_schedule:25, apply_conditioning:46, SinusoidalPosEmb:112,
ResidualTemporalBlock:290, DiffusionUNet1d:342, TemporalValue:562).
Re-designed: GroupNorm+Mish conv blocks assembled directly (no
einops/rearrange layers), attention dropped from the hot path — on MI355X
the [B, C, T] temporal convs map to MIOpen 1-D convs and the per-block
time-embedding add is fused by the eager allocator anyway.
"""
import math
from typing import List, Optional

import torch
import torch.nn as nn


def extract(a: torch.Tensor, t: torch.Tensor, x_shape) -> torch.Tensor:
    """Gather per-timestep schedule coefficients and broadcast to x_shape."""
    b = t.shape[0]
    out = a.gather(-1, t)
    return out.reshape(b, *((1, ) * (len(x_shape) - 1)))


def cosine_beta_schedule(timesteps: int, s: float = 0.008, dtype=torch.float32) -> torch.Tensor:
    steps = timesteps + 1
    x = torch.linspace(0, timesteps, steps)
    alphas_cumprod = torch.cos(((x / timesteps) + s) / (1 + s) * math.pi * 0.5) ** 2
    alphas_cumprod = alphas_cumprod / alphas_cumprod[0]
    betas = 1 - (alphas_cumprod[1:] / alphas_cumprod[:-1])
    return torch.clip(betas, 0, 0.999).to(dtype)


def apply_conditioning(x: torch.Tensor, conditions: dict, action_dim: int) -> torch.Tensor:
    """Pin known states into the trajectory: x[:, t, action_dim:] = state."""
    for t, val in conditions.items():
        x[:, t, action_dim:] = val.clone()
    return x


class SinusoidalPosEmb(nn.Module):

    def __init__(self, dim: int):
        super().__init__()
        self.dim = dim

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        half_dim = self.dim // 2
        emb = math.log(10000) / (half_dim - 1)
        emb = torch.exp(torch.arange(half_dim, device=x.device) * -emb)
        emb = x[:, None] * emb[None, :]
        return torch.cat([emb.sin(), emb.cos()], dim=-1)


class Conv1dBlock(nn.Module):
    """conv1d -> groupnorm -> mish."""

    def __init__(self, in_c: int, out_c: int, kernel: int, n_groups: int = 8):
        super().__init__()
        self.block = nn.Sequential(
            nn.Conv1d(in_c, out_c, kernel, padding=kernel // 2),
            nn.GroupNorm(n_groups, out_c),
            nn.Mish(),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.block(x)


class ResidualTemporalBlock(nn.Module):
    """Two conv blocks with a time-embedding shift between them."""

    def __init__(self, in_c: int, out_c: int, embed_dim: int, kernel: int = 5, mish: bool = True):
        super().__init__()
        self.blocks = nn.ModuleList([Conv1dBlock(in_c, out_c, kernel), Conv1dBlock(out_c, out_c, kernel)])
        act = nn.Mish() if mish else nn.SiLU()
        self.time_mlp = nn.Sequential(act, nn.Linear(embed_dim, out_c))
        self.residual_conv = nn.Conv1d(in_c, out_c, 1) if in_c != out_c else nn.Identity()

    def forward(self, x: torch.Tensor, t: torch.Tensor) -> torch.Tensor:
        out = self.blocks[0](x) + self.time_mlp(t).unsqueeze(-1)
        out = self.blocks[1](out)
        return out + self.residual_conv(x)


class DiffusionUNet1d(nn.Module):
    """Temporal UNet denoiser over [B, horizon, transition_dim] trajectories,
    optionally conditioned on returns (classifier-free guidance)."""

    def __init__(
        self,
        transition_dim: int,
        dim: int = 32,
        dim_mults: List[int] = [1, 2, 4, 8],
        returns_condition: bool = False,
        condition_dropout: float = 0.1,
        kernel_size: int = 5,
        **kwargs,
    ):
        super().__init__()
        dims = [transition_dim] + [dim * m for m in dim_mults]
        in_out = list(zip(dims[:-1], dims[1:]))
        time_dim = dim
        self.time_mlp = nn.Sequential(
            SinusoidalPosEmb(dim), nn.Linear(dim, dim * 4), nn.Mish(), nn.Linear(dim * 4, dim)
        )
        self.returns_condition = returns_condition
        self.condition_dropout = condition_dropout
        embed_dim = time_dim
        if returns_condition:
            self.returns_mlp = nn.Sequential(
                nn.Linear(1, dim), nn.Mish(), nn.Linear(dim, dim * 4), nn.Mish(), nn.Linear(dim * 4, dim)
            )
            embed_dim = 2 * time_dim

        self.downs = nn.ModuleList()
        self.ups = nn.ModuleList()
        n = len(in_out)
        for i, (c_in, c_out) in enumerate(in_out):
            last = i >= n - 1
            self.downs.append(nn.ModuleList([
                ResidualTemporalBlock(c_in, c_out, embed_dim, kernel_size),
                ResidualTemporalBlock(c_out, c_out, embed_dim, kernel_size),
                nn.Conv1d(c_out, c_out, 3, 2, 1) if not last else nn.Identity(),
            ]))
        mid_dim = dims[-1]
        self.mid_block1 = ResidualTemporalBlock(mid_dim, mid_dim, embed_dim, kernel_size)
        self.mid_block2 = ResidualTemporalBlock(mid_dim, mid_dim, embed_dim, kernel_size)
        for i, (c_in, c_out) in enumerate(reversed(in_out[1:])):
            # every downsample needs a mirroring upsample so the output
            # horizon matches the input trajectory length
            self.ups.append(nn.ModuleList([
                ResidualTemporalBlock(c_out * 2, c_in, embed_dim, kernel_size),
                ResidualTemporalBlock(c_in, c_in, embed_dim, kernel_size),
                nn.ConvTranspose1d(c_in, c_in, 4, 2, 1),
            ]))
        self.final_conv = nn.Sequential(
            Conv1dBlock(dim, dim, kernel_size), nn.Conv1d(dim, transition_dim, 1)
        )

    def forward(self, x: torch.Tensor, cond, time: torch.Tensor, returns: Optional[torch.Tensor] = None,
                use_dropout: bool = True, force_dropout: bool = False) -> torch.Tensor:
        x = x.transpose(1, 2)  # [B, T, D] -> [B, D, T]
        t = self.time_mlp(time)
        if self.returns_condition and returns is not None:
            r = self.returns_mlp(returns)
            if use_dropout and self.training:
                mask = (torch.rand(r.shape[0], 1, device=r.device) > self.condition_dropout).float()
                r = r * mask
            if force_dropout:
                r = r * 0
            t = torch.cat([t, r], dim=-1)
        elif self.returns_condition:
            t = torch.cat([t, torch.zeros_like(t)], dim=-1)
        h = []
        for block1, block2, down in self.downs:
            x = block1(x, t)
            x = block2(x, t)
            h.append(x)
            x = down(x)
        x = self.mid_block1(x, t)
        x = self.mid_block2(x, t)
        for block1, block2, up in self.ups:
            x = torch.cat([x, h.pop()], dim=1)
            x = block1(x, t)
            x = block2(x, t)
            x = up(x)
        x = self.final_conv(x)
        return x.transpose(1, 2)

    def get_pred(self, x, cond, time, returns=None, use_dropout: bool = True, force_dropout: bool = False):
        return self.forward(x, cond, time, returns, use_dropout, force_dropout)


class TemporalValue(nn.Module):
    """Trajectory -> scalar value (diffusion guidance critic)."""

    def __init__(
        self,
        horizon: int,
        transition_dim: int,
        dim: int = 32,
        dim_mults: List[int] = [1, 2, 4, 8],
        out_dim: int = 1,
        kernel_size: int = 5,
        **kwargs,
    ):
        super().__init__()
        dims = [transition_dim] + [dim * m for m in dim_mults]
        in_out = list(zip(dims[:-1], dims[1:]))
        time_dim = dim
        self.time_mlp = nn.Sequential(
            SinusoidalPosEmb(dim), nn.Linear(dim, dim * 4), nn.Mish(), nn.Linear(dim * 4, dim)
        )
        self.blocks = nn.ModuleList()
        h = horizon
        for c_in, c_out in in_out:
            self.blocks.append(nn.ModuleList([
                ResidualTemporalBlock(c_in, c_out, time_dim, kernel_size),
                ResidualTemporalBlock(c_out, c_out, time_dim, kernel_size),
                nn.Conv1d(c_out, c_out, 3, 2, 1),
            ]))
            h = (h + 1) // 2
        mid_dim = dims[-1]
        self.mid_block = ResidualTemporalBlock(mid_dim, mid_dim, time_dim, kernel_size)
        self.final = nn.Sequential(
            nn.Linear(mid_dim * max(h, 1) + time_dim, mid_dim // 2), nn.Mish(), nn.Linear(mid_dim // 2, out_dim)
        )

    def forward(self, x: torch.Tensor, cond, time: torch.Tensor, *args) -> torch.Tensor:
        x = x.transpose(1, 2)
        t = self.time_mlp(time)
        for block1, block2, down in self.blocks:
            x = block1(x, t)
            x = block2(x, t)
            x = down(x)
        x = self.mid_block(x, t)
        x = x.flatten(1)
        return self.final(torch.cat([x, t], dim=-1))
