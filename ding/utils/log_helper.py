"""Logging: python logger + a tensorboard-compatible scalar writer.

The offline image has no tensorboard package, so ``SummaryWriter`` here is a
self-contained JSONL event writer exposing the tb API subset DI-engine uses
(add_scalar/add_scalars/add_histogram/add_video/close/flush). Files land in
``<exp>/log/`` as ``events.jsonl`` — human-greppable and machine-parseable.

Parity: reference ding/utils/log_helper.py:15 build_logger,
ding/utils/log_writer_helper.py:13 DistributedWriter.
"""
import json
import logging
import numbers
import os
import sys
import threading
import time
from typing import Optional, Tuple

import numpy as np


def _to_scalar(v):
    if isinstance(v, numbers.Number):
        return float(v)
    try:
        import torch
        if isinstance(v, torch.Tensor):
            return float(v.detach().float().mean().item())
    except ImportError:
        pass
    if isinstance(v, np.ndarray):
        return float(v.mean())
    return None


class SummaryWriter:
    """JSONL scalar/histogram writer with the tensorboard API surface."""

    def __init__(self, log_dir: str = "./log", **kwargs):
        self.log_dir = log_dir
        os.makedirs(log_dir, exist_ok=True)
        self._path = os.path.join(log_dir, "events.jsonl")
        self._fh = open(self._path, "a", buffering=1)
        self._lock = threading.Lock()

    def _write(self, kind: str, tag: str, value, global_step: Optional[int]):
        rec = {"t": time.time(), "kind": kind, "tag": tag, "value": value, "step": global_step}
        with self._lock:
            self._fh.write(json.dumps(rec) + "\n")

    def add_scalar(self, tag: str, value, global_step: Optional[int] = None, **kw):
        s = _to_scalar(value)
        if s is not None:
            self._write("scalar", tag, s, global_step)

    def add_scalars(self, main_tag: str, tag_scalar_dict: dict, global_step: Optional[int] = None, **kw):
        for k, v in tag_scalar_dict.items():
            self.add_scalar(f"{main_tag}/{k}", v, global_step)

    def add_histogram(self, tag: str, values, global_step: Optional[int] = None, **kw):
        try:
            arr = np.asarray(
                values.detach().cpu().numpy() if hasattr(values, "detach") else values, dtype=np.float64
            ).ravel()
            summary = {
                "min": float(arr.min()), "max": float(arr.max()),
                "mean": float(arr.mean()), "std": float(arr.std()), "n": int(arr.size),
            }
        except Exception:
            return
        self._write("histogram", tag, summary, global_step)

    def add_text(self, tag: str, text: str, global_step: Optional[int] = None, **kw):
        self._write("text", tag, str(text), global_step)

    def add_video(self, *args, **kw):  # no-op without media stack
        pass

    def add_image(self, *args, **kw):
        pass

    def flush(self):
        with self._lock:
            self._fh.flush()

    def close(self):
        try:
            self.flush()
            self._fh.close()
        except Exception:
            pass


class DistributedWriter(SummaryWriter):
    """Rank-aware writer: only the designated writer rank persists records.

    In RCCL DP mode each process constructs one of these; non-zero ranks drop
    writes locally (cheap, no event-bus forwarding needed since every learner
    rank computes the same reduced metrics). In event-bus parallel mode the
    ``Parallel`` router plugs itself in via :meth:`plugin` and forwards writes
    to the labelled writer node (reference log_writer_helper.py:66-128).
    """

    _instance = None
    _default_key = None

    def __init__(self, log_dir: str = "./log", **kwargs):
        super().__init__(log_dir, **kwargs)
        self._is_writer = True
        self._router = None

    def plugin(self, router, is_writer: bool) -> "DistributedWriter":
        self._router = router
        self._is_writer = is_writer
        if router is not None and is_writer:
            router.on("distributed_writer", self._on_remote)
        return self

    def _on_remote(self, kind: str, tag: str, value, step):
        getattr(super(), "add_" + kind, lambda *a, **k: None)(tag, value, step)

    def _write(self, kind, tag, value, step):
        if self._is_writer:
            super()._write(kind, tag, value, step)
        elif self._router is not None:
            try:
                self._router.emit("distributed_writer", "scalar", tag, value, step, only_remote=True)
            except Exception:
                pass

    @classmethod
    def get_instance(cls, *args, **kwargs) -> Optional["DistributedWriter"]:
        if args or kwargs:
            cls._instance = cls(*args, **kwargs)
        return cls._instance


def build_logger(
    path: str = "./log",
    name: Optional[str] = None,
    need_tb: bool = True,
    need_text: bool = True,
    text_level=logging.INFO,
) -> Tuple[Optional[logging.Logger], Optional[SummaryWriter]]:
    """Create a text logger writing to ``<path>/<name>_logger.txt`` (+stderr)
    and a scalar writer under ``<path>``."""
    name = name or "default"
    logger, tb = None, None
    if need_text:
        logger = logging.getLogger(name)
        logger.setLevel(text_level)
        if not logger.handlers:
            os.makedirs(path, exist_ok=True)
            fh = logging.FileHandler(os.path.join(path, f"{name}_logger.txt"))
            fmt = logging.Formatter("[%(asctime)s][%(name)s][%(levelname)s] %(message)s")
            fh.setFormatter(fmt)
            sh = logging.StreamHandler(sys.stderr)
            sh.setFormatter(fmt)
            logger.addHandler(fh)
            logger.addHandler(sh)
        logger.propagate = False
    if need_tb:
        tb = SummaryWriter(path)
    return logger, tb


def pretty_print(data: dict, direct_print: bool = True) -> str:
    text = json.dumps(data, indent=2, default=str)
    if direct_print:
        print(text)
    return text


class LoggerFactory:
    """Factory view over build_logger: plain-logging creation plus the
    tabulate helpers (reference log_helper.py:64)."""

    @classmethod
    def create_logger(cls, path, name: str = 'default', level=logging.INFO):
        import os
        os.makedirs(path, exist_ok=True)
        logger = logging.getLogger(name)
        logger.setLevel(level)
        if not any(isinstance(h, logging.FileHandler) for h in logger.handlers):
            fh = logging.FileHandler(os.path.join(path, f'{name}_logger.txt'))
            fh.setFormatter(logging.Formatter('[%(asctime)s][%(levelname)s] %(message)s'))
            logger.addHandler(fh)
        return logger

    @staticmethod
    def get_tabulate_vars(variables: dict) -> str:
        from tabulate import tabulate
        return tabulate([[k, v] for k, v in variables.items()], headers=['Name', 'Value'], tablefmt='grid')

    @staticmethod
    def get_tabulate_vars_hor(variables: dict) -> str:
        from tabulate import tabulate
        keys, vals = list(variables.keys()), list(variables.values())
        return tabulate([vals], headers=keys, tablefmt='grid')
