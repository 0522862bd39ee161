"""Buffer ABC with a composable middleware chain.

Parity: reference ding/data/buffer/buffer.py (Buffer:56, apply_middleware,
BufferedData). Middleware are callables intercepting (action, chain, *args):
they may rewrite arguments, call ``chain`` to continue, and post-process.
"""
import copy
import uuid
from abc import ABC, abstractmethod
from dataclasses import dataclass, field
from typing import Any, Callable, List, Optional, Union


@dataclass
class BufferedData:
    data: Any
    index: str
    meta: dict = field(default_factory=dict)


def apply_middleware(func_name: str):

    def wrap(f: Callable) -> Callable:

        def handler(buffer, *args, **kwargs):
            """Walk the middleware chain outermost-first, innermost = the
            buffer method itself."""

            def wrap_handler(middleware: List[Callable], *args, **kwargs):
                if len(middleware) == 0:
                    return f(buffer, *args, **kwargs)

                def chain(*args, **kwargs):
                    return wrap_handler(middleware[1:], *args, **kwargs)

                return middleware[0](func_name, chain, *args, **kwargs)

            return wrap_handler(buffer.middleware, *args, **kwargs)

        return handler

    return wrap


class Buffer(ABC):

    def __init__(self, size: int):
        self.size = size
        self.middleware: List[Callable] = []

    @abstractmethod
    def push(self, data: Any, meta: Optional[dict] = None) -> BufferedData:
        raise NotImplementedError

    @abstractmethod
    def sample(self, size: Optional[int] = None, **kwargs) -> List[BufferedData]:
        raise NotImplementedError

    @abstractmethod
    def update(self, index: str, data: Optional[Any] = None, meta: Optional[dict] = None) -> bool:
        raise NotImplementedError

    @abstractmethod
    def delete(self, index: str) -> bool:
        raise NotImplementedError

    @abstractmethod
    def count(self) -> int:
        raise NotImplementedError

    @abstractmethod
    def clear(self) -> None:
        raise NotImplementedError

    @abstractmethod
    def get(self, idx: int) -> BufferedData:
        raise NotImplementedError

    def use(self, func: Callable) -> "Buffer":
        """Append a middleware; returns self for chaining."""
        self.middleware.append(func)
        return self

    def view(self) -> "Buffer":
        """Shallow copy sharing storage but with an independent middleware list."""
        buffer = copy.copy(self)
        buffer.middleware = list(self.middleware)
        return buffer

    def __copy__(self):
        cls = self.__class__
        new = cls.__new__(cls)
        new.__dict__.update(self.__dict__)
        return new

    def save_data(self, file_name: str):
        from ding.utils import save_file
        save_file(file_name, self.export_data())

    def load_data(self, file_name: str):
        from ding.utils import read_file
        self.import_data(read_file(file_name))

    def export_data(self) -> List[BufferedData]:
        raise NotImplementedError

    def import_data(self, data: List[BufferedData]) -> None:
        raise NotImplementedError


def fastcopy_uuid() -> str:
    return uuid.uuid4().hex
