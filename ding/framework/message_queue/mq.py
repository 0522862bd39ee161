"""Message-queue transport ABC.

Parity: reference ding/framework/message_queue/mq.py:4. Backends: tcp
(stdlib, default offline), nng/redis kept as optional named backends that
raise with a clear message when their client libs are absent.
"""
from abc import ABC, abstractmethod
from typing import Optional, Tuple


class MQ(ABC):

    def __init__(self, *args, **kwargs) -> None:
        pass

    @abstractmethod
    def listen(self) -> None:
        raise NotImplementedError

    @abstractmethod
    def publish(self, topic: str, data: bytes) -> None:
        raise NotImplementedError

    @abstractmethod
    def subscribe(self, topic: str) -> None:
        raise NotImplementedError

    @abstractmethod
    def unsubscribe(self, topic: str) -> None:
        raise NotImplementedError

    @abstractmethod
    def recv(self) -> Tuple[str, bytes]:
        raise NotImplementedError

    def stop(self) -> None:
        pass
