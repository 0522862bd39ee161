"""Ship model state_dicts through storage objects in the background.

Parity: reference ding/data/model_loader.py (ModelLoader:28, FileModelLoader).
"""
import logging
import os
import threading
import time
import uuid
from abc import ABC, abstractmethod
from typing import Any, Callable, Optional

import torch

from .storage import FileStorage, Storage


class ModelLoader(ABC):

    def __init__(self, model: torch.nn.Module):
        self._model = model
        self._send_callback: Optional[Callable] = None

    def start(self):
        pass

    @abstractmethod
    def save(self, callback: Callable) -> None:
        """Serialize the current state_dict asynchronously; call callback(storage)."""
        raise NotImplementedError

    @abstractmethod
    def load(self, storage: Storage) -> dict:
        raise NotImplementedError

    def shutdown(self):
        pass


class FileModelLoader(ModelLoader):

    def __init__(self, model: torch.nn.Module, dirname: str, ttl: int = 20):
        super().__init__(model)
        self._dirname = dirname
        self._ttl = ttl
        self._files = []
        self._thread: Optional[threading.Thread] = None

    def save(self, callback: Callable) -> None:
        def _save():
            os.makedirs(self._dirname, exist_ok=True)
            path = os.path.join(self._dirname, f"model_{uuid.uuid4().hex}.pth.tar")
            storage = FileStorage(path)
            state = {k: v.detach().cpu() for k, v in self._model.state_dict().items()}
            storage.save(state)
            self._files.append((time.time(), path))
            self._cleanup()
            callback(storage)

        self._thread = threading.Thread(target=_save, daemon=True)
        self._thread.start()

    def load(self, storage: Storage) -> dict:
        return storage.load()

    def _cleanup(self):
        now = time.time()
        keep = []
        for ts, path in self._files:
            if now - ts > self._ttl:
                try:
                    os.remove(path)
                except OSError:
                    pass
            else:
                keep.append((ts, path))
        self._files = keep
