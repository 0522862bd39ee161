"""Shared-memory tensor buffers for zero-copy obs transfer from env
subprocesses.

Parity: reference ding/data/shm_buffer.py (ShmBuffer:22, ShmBufferContainer:81).
"""
import ctypes
import multiprocessing as mp
from typing import Any, Dict, Optional, Tuple, Union

import numpy as np
import torch

_NP_TO_CTYPE = {
    np.dtype(np.float32): ctypes.c_float,
    np.dtype(np.float64): ctypes.c_double,
    np.dtype(np.int32): ctypes.c_int32,
    np.dtype(np.int64): ctypes.c_int64,
    np.dtype(np.uint8): ctypes.c_uint8,
    np.dtype(np.bool_): ctypes.c_bool,
}


class ShmBuffer:
    """One fixed-shape array in a multiprocessing.Array; fill()/get() copy
    in/out without pickling."""

    def __init__(self, dtype: Union[type, np.dtype], shape: Tuple[int, ...], copy_on_get: bool = True):
        self.dtype = np.dtype(dtype)
        self.shape = shape
        self.copy_on_get = copy_on_get
        size = int(np.prod(shape))
        self.buffer = mp.Array(_NP_TO_CTYPE[self.dtype], size, lock=False)

    def fill(self, src_arr: np.ndarray) -> None:
        assert src_arr.dtype == self.dtype and src_arr.shape == self.shape, \
            f"{src_arr.dtype}/{src_arr.shape} vs {self.dtype}/{self.shape}"
        dst = np.frombuffer(self.buffer, dtype=self.dtype).reshape(self.shape)
        np.copyto(dst, src_arr)

    def get(self) -> np.ndarray:
        arr = np.frombuffer(self.buffer, dtype=self.dtype).reshape(self.shape)
        return arr.copy() if self.copy_on_get else arr


class ShmBufferContainer:
    """Nested dict of ShmBuffers keyed like the obs structure."""

    def __init__(self, dtype: Any, shape: Union[Dict[str, tuple], tuple], copy_on_get: bool = True):
        if isinstance(shape, dict):
            self._data = {k: ShmBufferContainer(dtype[k] if isinstance(dtype, dict) else dtype, v, copy_on_get)
                          for k, v in shape.items()}
            self._is_dict = True
        else:
            self._data = ShmBuffer(dtype, shape, copy_on_get)
            self._is_dict = False

    def fill(self, src: Any) -> None:
        if self._is_dict:
            for k in self._data:
                self._data[k].fill(src[k])
        else:
            self._data.fill(np.ascontiguousarray(src))

    def get(self) -> Any:
        if self._is_dict:
            return {k: v.get() for k, v in self._data.items()}
        return self._data.get()
