"""File IO for checkpoints/data payloads.

Parity: reference ding/utils/file_helper.py:24-111 (local/ceph/redis/memcached
backends). Only the local filesystem backend is real here; the others raise
with a clear message (no such services offline). Checkpoint format is
preserved: torch.save of a dict to ``*.pth.tar`` paths.
"""
import os
import pickle
from typing import Any

import torch


def read_file(path: str, fs_type: str = "normal", use_lock: bool = False) -> Any:
    if fs_type != "normal":
        raise NotImplementedError(f"fs_type '{fs_type}' backend not available offline")
    if use_lock:
        from .lock_helper import FcntlContext
        with FcntlContext(path + ".lock"):
            return torch.load(path, map_location="cpu", weights_only=False)
    return torch.load(path, map_location="cpu", weights_only=False)


def save_file(path: str, data: Any, fs_type: str = "normal", use_lock: bool = False) -> None:
    if fs_type != "normal":
        raise NotImplementedError(f"fs_type '{fs_type}' backend not available offline")
    d = os.path.dirname(path)
    if d:
        os.makedirs(d, exist_ok=True)
    if use_lock:
        from .lock_helper import FcntlContext
        with FcntlContext(path + ".lock"):
            torch.save(data, path)
    else:
        torch.save(data, path)


def remove_file(path: str, fs_type: str = "normal") -> None:
    try:
        os.remove(path)
    except FileNotFoundError:
        pass


def read_from_file(path: str) -> Any:
    with open(path, "rb") as f:
        return pickle.load(f)


def save_to_file(path: str, data: Any) -> None:
    d = os.path.dirname(path)
    if d:
        os.makedirs(d, exist_ok=True)
    with open(path, "wb") as f:
        pickle.dump(data, f, protocol=pickle.HIGHEST_PROTOCOL)
