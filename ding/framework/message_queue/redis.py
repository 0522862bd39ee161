"""Redis pub/sub MQ backend — optional; raises a clear error when redis-py
is not installed.

Parity: reference ding/framework/message_queue/redis.py (RedisMQ:12).
"""
from typing import Optional, Tuple

from .mq import MQ


class RedisMQ(MQ):

    def __init__(self, redis_host: str = '127.0.0.1', redis_port: int = 6379, **kwargs):
        try:
            import redis
        except ImportError as e:
            raise ImportError(
                "RedisMQ requires the redis client, which is not available in this "
                "offline build; use mq_type='tcp' (default) instead"
            ) from e
        self._client = redis.Redis(host=redis_host, port=int(redis_port), db=kwargs.get('db', 0))
        self._pubsub = self._client.pubsub()

    def listen(self) -> None:
        pass

    def publish(self, topic: str, data: bytes) -> None:
        self._client.publish(topic, data)

    def subscribe(self, topic: str) -> None:
        self._pubsub.subscribe(topic)

    def unsubscribe(self, topic: str) -> None:
        self._pubsub.unsubscribe(topic)

    def recv(self) -> Tuple[str, bytes]:
        while True:
            msg = self._pubsub.get_message(ignore_subscribe_messages=True, timeout=1.0)
            if msg and msg.get('type') == 'message':
                ch = msg['channel']
                return (ch.decode() if isinstance(ch, bytes) else ch), msg['data']

    def stop(self) -> None:
        try:
            self._pubsub.close()
            self._client.close()
        except Exception:
            pass
