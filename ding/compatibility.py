"""Torch feature gates. The MI355X build requires PyTorch-ROCm >= 2.x, so
the reference's historical version checks (ding/compatibility.py) are all
identically True here; they exist so user code importing them keeps working.
"""
import torch


def _ver() -> tuple:
    return tuple(int(x) for x in torch.__version__.split('+')[0].split('.')[:2])


def torch_ge_131() -> bool:
    return _ver() >= (1, 3)


def torch_ge_180() -> bool:
    return _ver() >= (1, 8)
