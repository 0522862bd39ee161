"""NNG (pynng) MQ backend — optional; raises a clear error when pynng is not
installed (this offline build defaults to the stdlib TCP backend).

Parity: reference ding/framework/message_queue/nng.py (NNGMQ:12).
"""
from typing import Optional, Tuple

from .mq import MQ


class NNGMQ(MQ):

    def __init__(self, listen_to: str, attach_to: Optional[list] = None, **kwargs):
        try:
            from pynng import Bus0
        except ImportError as e:
            raise ImportError(
                "NNGMQ requires pynng, which is not available in this offline build; "
                "use mq_type='tcp' (default) instead"
            ) from e
        self.listen_to = listen_to
        self.attach_to = attach_to or []
        self._sock = Bus0()
        self._running = False

    def listen(self) -> None:
        self._sock.listen(self.listen_to)
        import time
        time.sleep(0.1)
        for contact in self.attach_to:
            self._sock.dial(contact)
        self._running = True

    def publish(self, topic: str, data: bytes) -> None:
        assert self._running
        self._sock.send(topic.encode() + b'::' + data)

    def subscribe(self, topic: str) -> None:
        pass  # bus topology broadcasts; topic filter happens on recv

    def unsubscribe(self, topic: str) -> None:
        pass

    def recv(self) -> Tuple[str, bytes]:
        while True:
            msg = self._sock.recv()
            if b'::' in msg:
                topic, payload = msg.split(b'::', 1)
                return topic.decode(), payload

    def stop(self) -> None:
        if self._running:
            self._sock.close()
            self._running = False
