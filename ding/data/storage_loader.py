"""Async storage (de)serialization used by ContextExchanger to move big
trajectory payloads off the event-loop thread.

Parity: reference ding/data/storage_loader.py (StorageLoader:30,
FileStorageLoader). This build uses a worker thread (payloads are already
shared-memory-friendly numpy/torch buffers; a thread avoids the extra
process hop the reference needs for its pickle-heavy path).
"""
import os
import queue
import threading
import time
import uuid
from abc import ABC, abstractmethod
from typing import Any, Callable, Optional

from .storage import FileStorage, Storage


class StorageLoader(ABC):

    @abstractmethod
    def save(self, obj: Any) -> Storage:
        raise NotImplementedError

    @abstractmethod
    def load(self, storage: Storage, callback: Callable) -> None:
        raise NotImplementedError

    @abstractmethod
    def shutdown(self) -> None:
        raise NotImplementedError


class FileStorageLoader(StorageLoader):

    def __init__(self, dirname: str, ttl: int = 20):
        self._dirname = dirname
        self._ttl = ttl
        self._files = []
        self._load_queue: queue.Queue = queue.Queue()
        self._end = False
        self._worker: Optional[threading.Thread] = None

    def save(self, obj: Any) -> FileStorage:
        os.makedirs(self._dirname, exist_ok=True)
        path = os.path.join(self._dirname, f"{uuid.uuid4().hex}.pkl")
        storage = FileStorage(path)
        storage.save(obj)
        self._files.append((time.time(), path))
        self._cleanup()
        return storage

    def load(self, storage: Storage, callback: Callable) -> None:
        if self._worker is None:
            self._worker = threading.Thread(target=self._loop, daemon=True)
            self._worker.start()
        self._load_queue.put((storage, callback))

    def _loop(self):
        while not self._end:
            try:
                storage, callback = self._load_queue.get(timeout=0.5)
            except queue.Empty:
                continue
            try:
                callback(storage.load())
            except FileNotFoundError:
                pass

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

    def shutdown(self) -> None:
        self._end = True
        if self._worker is not None:
            self._worker.join(timeout=1)
