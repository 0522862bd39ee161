"""Data-plane middleware: pushers and fetchers between collector and buffer.

Parity: reference ding/framework/middleware/functional/data_processor.py
(data_pusher:17, offpolicy_data_fetcher:91, offline_data_fetcher:186,
buffer_saver:55).
"""
import logging
from typing import Callable, List, Optional, Union

import torch

from ding.data import Buffer
from ding.data.buffer.middleware import PriorityExperienceReplay
from ding.utils import EasyDict
from ...context import OnlineRLContext, OfflineRLContext

logger = logging.getLogger('ding')


def data_pusher(cfg: EasyDict, buffer_: Buffer, group_by_env: Optional[bool] = None) -> Callable:
    """Push ctx.trajectories (or episodes) into the buffer."""

    def _push(ctx: OnlineRLContext):
        if ctx.trajectories is not None:
            if group_by_env:
                for t in ctx.trajectories:
                    buffer_.push(t, {'env': t.env_data_id.item()})
            else:
                for t in ctx.trajectories:
                    buffer_.push(t)
            ctx.trajectories = None
        elif ctx.episodes is not None:
            for episode in ctx.episodes:
                buffer_.push(episode)
            ctx.episodes = None
        else:
            raise RuntimeError("no trajectories or episodes to push")

    return _push


def buffer_saver(cfg: EasyDict, buffer_: Buffer, every_envstep: int = 1000, replace: bool = False) -> Callable:
    """Periodically persist the buffer to <exp_name>/replay_buffer."""
    last = [0]

    def _save(ctx: OnlineRLContext):
        if ctx.env_step - last[0] >= every_envstep:
            last[0] = ctx.env_step
            suffix = "data_latest" if replace else f"data_envstep_{ctx.env_step}"
            buffer_.save_data(f"{cfg.exp_name}/replay_buffer/{suffix}.hkl")

    return _save


def offpolicy_data_fetcher(
    cfg: EasyDict,
    buffer_: Union[Buffer, List[Buffer], dict],
    data_shortage_warning: bool = False,
) -> Callable:
    """Sample a train batch into ctx.train_data; stream priority updates back
    to the buffer on the backward pass (generator middleware)."""

    def _fetch(ctx: OnlineRLContext):
        try:
            batch_size = cfg.policy.learn.batch_size
            if isinstance(buffer_, Buffer):
                buffered = buffer_.sample(batch_size)
            elif isinstance(buffer_, list):  # sample proportionally
                buffered = []
                for b, ratio in buffer_:
                    buffered.extend(b.sample(int(batch_size * ratio)))
            elif isinstance(buffer_, dict):
                buffered = {k: b.sample(batch_size) for k, b in buffer_.items()}
            else:
                raise TypeError(type(buffer_))
        except (ValueError, AssertionError):
            if data_shortage_warning:
                logger.warning("replay buffer shortage: skip this training round")
            ctx.train_data = None
            return

        if isinstance(buffered, dict):
            index = {k: [d.index for d in v] for k, v in buffered.items()}
            meta = {k: [d.meta for d in v] for k, v in buffered.items()}
            data = {k: [d.data for d in v] for k, v in buffered.items()}
            for k in data:
                for d, m in zip(data[k], meta[k]):
                    if 'priority_IS' in m:
                        d['priority_IS'] = torch.as_tensor([m['priority_IS']])
        else:
            index = [d.index for d in buffered]
            meta = [d.meta for d in buffered]
            data = [d.data for d in buffered]
            for d, m in zip(data, meta):
                if isinstance(d, dict) and 'priority_IS' in m:
                    d['priority_IS'] = torch.as_tensor([m['priority_IS']])
        ctx.train_data = data

        yield

        # backward: write updated priorities
        if ctx.train_output is not None:
            out = ctx.train_output
            if isinstance(out, (list, tuple)) and len(out) > 0:
                out = out[-1]
            if isinstance(out, dict) and 'priority' in out and out['priority'] is not None and \
                    isinstance(buffer_, Buffer):
                for idx, prio in zip(index, out['priority']):
                    buffer_.update(idx, data=None, meta={'priority': float(prio)})

    return _fetch


def offline_data_fetcher(cfg: EasyDict, dataset, collate_fn=None) -> Callable:
    """Epoch-wise minibatch iterator over an offline dataset."""
    from torch.utils.data import DataLoader

    def _collate(batch):
        return list(batch) if collate_fn is None else collate_fn(batch)

    dataloader = DataLoader(
        dataset, batch_size=cfg.policy.learn.batch_size, shuffle=True, collate_fn=_collate
    )

    def produce():
        while True:
            for batch in dataloader:
                yield batch

    stream = produce()

    def _fetch(ctx: OfflineRLContext):
        ctx.train_data = next(stream)
        ctx.train_epoch = getattr(ctx, 'train_epoch', 0)

    return _fetch


def offline_data_fetcher_from_mem(cfg: EasyDict, dataset) -> Callable:
    import random

    def _fetch(ctx: OfflineRLContext):
        idx = random.sample(range(len(dataset)), cfg.policy.learn.batch_size)
        ctx.train_data = [dataset[i] for i in idx]

    return _fetch


def sqil_data_pusher(cfg: EasyDict, buffer_: Buffer, expert: bool) -> Callable:
    """SQIL: label pushed transitions with constant reward 1 (expert) / 0."""

    def _push(ctx: OnlineRLContext):
        for t in ctx.trajectories:
            t = EasyDict(t)
            t.reward = torch.ones_like(t.reward) if expert else torch.zeros_like(t.reward)
            buffer_.push(t)
        ctx.trajectories = None

    return _push
