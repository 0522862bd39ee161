"""Same-node GPU trajectory exchange middleware.

Replaces pickle-over-TCP trajectory shipping (ContextExchanger) for
co-located actor/learner ranks that share a torch.distributed group: the
actor middleware pushes ``ctx.trajectories``/``ctx.train_data`` through
``ding.data.TrajectoryShipper`` (flat per-dtype ``dist.send`` — RCCL over
xGMI GPU->GPU on a real node, gloo on CPU), the learner middleware
receives into ``ctx.train_data``. Header-only pickle; tensor payloads never
touch the host on the nccl backend.

Reference counterpart: opendilab/DI-engine ding/data/storage_loader.py
FileStorage + shm hand-off.
"""
from typing import Callable, Optional

import torch
import torch.distributed as dist

from ding.data import TrajectoryShipper
from ding.utils.data import default_collate


def gpu_trajectory_sender(dst: int, collate: bool = True, device: Optional[str] = None) -> Callable:
    """Actor-side middleware: ship this iteration's trajectories to ``dst``.

    With ``collate=True`` the list of transition dicts is collated into one
    tensor batch first, so the wire carries a handful of large messages
    (xGMI-friendly) instead of thousands of per-transition tensors.
    """
    assert dist.is_available() and dist.is_initialized(), "init_process_group first"
    shipper = TrajectoryShipper()

    def _send(ctx):
        batch = getattr(ctx, 'trajectories', None) or getattr(ctx, 'train_data', None)
        if not batch:
            return
        payload = batch
        if collate and isinstance(batch, list) and isinstance(batch[0], dict):
            payload = default_collate([dict(d) for d in batch], cat_1dim=True)
        if device is not None:
            from ding.torch_utils import to_device
            payload = to_device(payload, device)
        shipper.send({'batch': payload, 'env_step': int(getattr(ctx, 'env_step', 0))}, dst=dst)

    return _send


def gpu_trajectory_receiver(src: int, device: Optional[str] = None) -> Callable:
    """Learner-side middleware: receive a batch from ``src`` into
    ``ctx.train_data`` (collated dict — policies with a collated-batch fast
    path, e.g. IMPALAPolicy, consume it without host re-collation)."""
    assert dist.is_available() and dist.is_initialized(), "init_process_group first"
    shipper = TrajectoryShipper()

    def _recv(ctx):
        msg = shipper.recv(src=src)
        batch = msg['batch']
        if device is not None:
            from ding.torch_utils import to_device
            batch = to_device(batch, device)
        ctx.train_data = batch
        if 'env_step' in msg:
            ctx.env_step = max(int(getattr(ctx, 'env_step', 0)), msg['env_step'])

    return _recv
