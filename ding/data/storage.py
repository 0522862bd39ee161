"""Storage objects: serialize payloads to a location, load them back.

Parity: reference ding/data/storage/storage.py + storage/file.py.
"""
import os
import pickle
import uuid
from abc import ABC, abstractmethod
from typing import Any


class Storage(ABC):

    def __init__(self, path: str):
        self.path = path

    @abstractmethod
    def save(self, data: Any) -> None:
        raise NotImplementedError

    @abstractmethod
    def load(self) -> Any:
        raise NotImplementedError


class FileStorage(Storage):

    def save(self, data: Any) -> None:
        d = os.path.dirname(self.path)
        if d:
            os.makedirs(d, exist_ok=True)
        with open(self.path, "wb") as f:
            pickle.dump(data, f, protocol=pickle.HIGHEST_PROTOCOL)

    def load(self) -> Any:
        with open(self.path, "rb") as f:
            return pickle.load(f)


class FileModelStorage(Storage):
    """Model-checkpoint storage on the filesystem (reference
    ding/data/storage/file.py:19)."""

    def save(self, state_dict: object) -> None:
        from ding.utils import save_file
        save_file(self.path, state_dict)

    def load(self) -> object:
        from ding.utils import read_file
        return read_file(self.path)
