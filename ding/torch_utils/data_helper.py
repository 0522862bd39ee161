"""Tensor/device conversion helpers and the async GPU fetcher.

Parity: reference ding/torch_utils/data_helper.py (to_device, to_tensor,
to_ndarray, to_list, same_shape, CudaFetcher:523). CudaFetcher prefetches
host batches onto the MI355X over a dedicated HIP stream (torch.cuda.Stream
is a hipStream on ROCm) so H2D copies overlap the learner's compute stream.
"""
import threading
import time
from collections.abc import Sequence
from typing import Any, Iterable, Optional

import numpy as np
import torch


def to_device(item: Any, device: str, ignore_keys: list = ()) -> Any:
    if isinstance(item, torch.nn.Module):
        return item.to(device)
    if isinstance(item, torch.Tensor):
        return item.to(device)
    if isinstance(item, dict):
        return {k: v if k in ignore_keys else to_device(v, device) for k, v in item.items()}
    if isinstance(item, tuple) and hasattr(item, "_fields"):
        return type(item)(*[to_device(v, device) for v in item])
    if isinstance(item, (list, tuple)):
        return type(item)(to_device(v, device) for v in item)
    if isinstance(item, (int, float, str, bool, bytes, np.ndarray, np.str_, type(None))):
        return item
    if isinstance(item, torch.distributions.Distribution):
        return item
    raise TypeError(f"to_device: unsupported type {type(item)}")


def to_dtype(item: Any, dtype) -> Any:
    if isinstance(item, torch.Tensor):
        return item.to(dtype)
    if isinstance(item, dict):
        return {k: to_dtype(v, dtype) for k, v in item.items()}
    if isinstance(item, Sequence) and not isinstance(item, str):
        return type(item)(to_dtype(v, dtype) for v in item)
    raise TypeError(f"to_dtype: unsupported type {type(item)}")


def to_tensor(item: Any, dtype: Optional[torch.dtype] = None, ignore_keys: list = (), transform_scalar: bool = True) -> Any:
    if isinstance(item, dict):
        return {k: v if k in ignore_keys else to_tensor(v, dtype, ignore_keys, transform_scalar) for k, v in item.items()}
    if isinstance(item, torch.Tensor):
        return item if dtype is None else item.to(dtype)
    if isinstance(item, np.ndarray):
        if dtype is None:
            if item.dtype == np.float64:
                return torch.from_numpy(item.astype(np.float32))
            return torch.from_numpy(item)
        return torch.from_numpy(item).to(dtype)
    if isinstance(item, bool) or isinstance(item, str):
        return item
    if isinstance(item, (list, tuple)):
        if len(item) == 0:
            return [] if isinstance(item, list) else ()
        if hasattr(item, "_fields"):  # namedtuple
            return type(item)(*[to_tensor(v, dtype) for v in item])
        if isinstance(item[0], str):
            return item  # text payloads (prompt policies) pass through
        if np.isscalar(item[0]):
            return torch.as_tensor(item, dtype=dtype if dtype is not None else torch.float32)
        return type(item)(to_tensor(v, dtype, ignore_keys, transform_scalar) for v in item)
    if np.isscalar(item):
        if transform_scalar:
            return torch.as_tensor(item, dtype=dtype if dtype is not None else (
                torch.int64 if isinstance(item, (int, np.integer)) else torch.float32
            ))
        return item
    if item is None:
        return None
    raise TypeError(f"to_tensor: unsupported type {type(item)}")


def to_ndarray(item: Any, dtype: Optional[np.dtype] = None) -> Any:
    if isinstance(item, dict):
        return {k: to_ndarray(v, dtype) for k, v in item.items()}
    if isinstance(item, torch.Tensor):
        arr = item.detach().cpu().numpy()
        return arr if dtype is None else arr.astype(dtype)
    if isinstance(item, np.ndarray):
        return item if dtype is None else item.astype(dtype)
    if isinstance(item, (list, tuple)):
        if len(item) == 0:
            return None
        if hasattr(item, "_fields"):
            return type(item)(*[to_ndarray(v, dtype) for v in item])
        if np.isscalar(item[0]):
            return np.asarray(item, dtype=dtype)
        return type(item)(to_ndarray(v, dtype) for v in item)
    if np.isscalar(item):
        return np.asarray(item, dtype=dtype)
    if item is None:
        return None
    raise TypeError(f"to_ndarray: unsupported type {type(item)}")


def to_list(item: Any) -> Any:
    if item is None:
        return None
    if isinstance(item, torch.Tensor):
        return item.tolist()
    if isinstance(item, np.ndarray):
        return item.tolist()
    if isinstance(item, dict):
        return {k: to_list(v) for k, v in item.items()}
    if isinstance(item, (list, tuple)):
        return [to_list(v) for v in item]
    if np.isscalar(item):
        return item
    raise TypeError(f"to_list: unsupported type {type(item)}")


def to_item(item: Any, ignore_error: bool = False) -> Any:
    """Tensors/arrays -> python scalars (for logging dicts)."""
    if isinstance(item, dict):
        return {k: to_item(v, ignore_error) for k, v in item.items()}
    if isinstance(item, (torch.Tensor, np.ndarray)):
        try:
            return item.item()
        except (ValueError, RuntimeError):
            if ignore_error:
                return None
            raise
    if isinstance(item, (list, tuple)):
        return type(item)(to_item(v, ignore_error) for v in item)
    return item


def same_shape(data: list) -> bool:
    assert isinstance(data, list)
    shapes = [t.shape for t in data]
    return len(set(shapes)) <= 1


def build_log_buffer():
    from collections import defaultdict
    return defaultdict(list)


def get_tensor_data(data: Any) -> Any:
    """Detach-and-share view of (nested) tensors (no copy)."""
    if isinstance(data, torch.Tensor):
        return data.data.clone()
    if isinstance(data, dict):
        return {k: get_tensor_data(v) for k, v in data.items()}
    if isinstance(data, (list, tuple)):
        return type(data)(get_tensor_data(v) for v in data)
    return data


def unsqueeze(data: Any, dim: int = 0) -> Any:
    if isinstance(data, torch.Tensor):
        return data.unsqueeze(dim)
    if isinstance(data, dict):
        return {k: unsqueeze(v, dim) for k, v in data.items()}
    raise TypeError(type(data))


def squeeze(data: Any, dim: int = 0) -> Any:
    if isinstance(data, torch.Tensor):
        return data.squeeze(dim)
    if isinstance(data, dict):
        return {k: squeeze(v, dim) for k, v in data.items()}
    raise TypeError(type(data))


def zeros_like(h: Any) -> Any:
    if isinstance(h, torch.Tensor):
        return torch.zeros_like(h)
    if isinstance(h, dict):
        return {k: zeros_like(v) for k, v in h.items()}
    if isinstance(h, (list, tuple)):
        return type(h)(zeros_like(v) for v in h)
    raise TypeError(type(h))


class CudaFetcher:
    """Background thread that moves batches from a host iterator to the GPU
    on a dedicated HIP copy stream, exposing a ready queue to the trainer."""

    def __init__(self, data_source: Iterable, device: str, queue_size: int = 4, sleep: float = 0.1):
        self._source = data_source
        self._device = device
        self._queue_size = queue_size
        self._sleep = sleep
        import queue
        self._queue = queue.Queue(maxsize=queue_size)
        self._end_flag = True
        self._stream: Optional[torch.cuda.Stream] = None
        self._thread: Optional[threading.Thread] = None

    def run(self) -> None:
        self._end_flag = False
        self._stream = torch.cuda.Stream(device=self._device)
        self._thread = threading.Thread(target=self._producer, daemon=True)
        self._thread.start()

    def close(self) -> None:
        self._end_flag = True

    def _producer(self) -> None:
        with torch.cuda.stream(self._stream):
            while not self._end_flag:
                if self._queue.full():
                    time.sleep(self._sleep)
                    continue
                try:
                    data = next(self._source)
                except StopIteration:
                    break
                data = to_device(data, self._device)
                # make the copy visible to the default compute stream
                torch.cuda.current_stream(self._device).synchronize()
                self._queue.put(data)

    def __next__(self) -> Any:
        return self._queue.get()

    def __iter__(self):
        return self


def tensor_to_list(item):
    """Recursively convert tensors to (nested) python lists, leaving
    non-tensor values untouched (reference data_helper.py:335)."""
    if item is None:
        return None
    if isinstance(item, torch.Tensor):
        return item.tolist()
    if isinstance(item, dict):
        return {k: tensor_to_list(v) for k, v in item.items()}
    if isinstance(item, (list, tuple)):
        return [tensor_to_list(v) for v in item]
    if isinstance(item, (int, float, str, bool)):
        return item
    raise TypeError(f"not supported item type: {type(item)}")


def get_null_data(template, num: int) -> list:
    """Padding transitions stamped null+done with zeroed reward, used to pad
    ragged episode batches (reference data_helper.py:696)."""
    import copy as _copy
    out = []
    for _ in range(num):
        data = _copy.deepcopy(template)
        data['null'] = True
        data['done'] = True
        data['reward'].zero_()
        out.append(data)
    return out
