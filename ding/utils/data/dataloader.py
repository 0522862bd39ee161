"""Async producer/consumer dataloader with GPU prefetch.

Parity: reference ding/utils/data/dataloader.py:15 (AsyncDataLoader —
worker procs collate batches, dedicated CUDA/HIP stream prefetch). Offline
build uses worker THREADS for collation (payloads are tensors already in
shared address space) + a dedicated HIP stream for H2D copies.
"""
import queue
import threading
from typing import Any, Callable, Iterable, Optional

import torch

from ding.torch_utils import to_device


class AsyncDataLoader:

    def __init__(
        self,
        data_source: Callable,
        batch_size: int,
        device: str = 'cpu',
        chunk_size: Optional[int] = None,
        collate_fn: Optional[Callable] = None,
        num_workers: int = 2,
        queue_maxsize: int = 4,
    ):
        """data_source(batch_size) -> list of sample dicts (or callables)."""
        self._source = data_source
        self._batch_size = batch_size
        self._device = device
        self._use_gpu = device.startswith('cuda') and torch.cuda.is_available()
        if collate_fn is None:
            from .collate_fn import default_collate
            collate_fn = default_collate
        self._collate = collate_fn
        self._queue: "queue.Queue" = queue.Queue(maxsize=queue_maxsize)
        self._end = False
        self._workers = [
            threading.Thread(target=self._produce, daemon=True) for _ in range(max(1, num_workers))
        ]
        self._stream = torch.cuda.Stream() if self._use_gpu else None
        for w in self._workers:
            w.start()

    def _produce(self):
        while not self._end:
            try:
                samples = self._source(self._batch_size)
                if callable(samples):
                    samples = samples()
                batch = self._collate(samples)
                if self._use_gpu:
                    with torch.cuda.stream(self._stream):
                        batch = to_device(batch, self._device)
                    self._stream.synchronize()
                self._queue.put(batch, timeout=5)
            except queue.Full:
                continue
            except Exception:
                if self._end:
                    break
                raise

    def __iter__(self):
        return self

    def __next__(self) -> Any:
        while not self._end:
            try:
                return self._queue.get(timeout=1)
            except queue.Empty:
                continue
        raise StopIteration

    def close(self):
        self._end = True
