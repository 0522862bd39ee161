"""Stdlib-TCP bus transport: every node binds one listening socket and dials
every peer (mesh) or the hub (star). Messages are length-prefixed
``topic::payload`` frames, matching the reference's nng Bus0 semantics
(ding/framework/message_queue/nng.py:12-49) without the external nng
dependency.
"""
import logging
import queue
import socket
import struct
import threading
import time
from typing import List, Optional, Tuple

from ding.utils import MQ_REGISTRY
from .mq import MQ

logger = logging.getLogger('ding')

_HEADER = struct.Struct("!I")


def _parse_addr(addr: str) -> Tuple[str, int]:
    # accepts "tcp://host:port" or "host:port"
    addr = addr.replace("tcp://", "")
    host, port = addr.rsplit(":", 1)
    return host, int(port)


@MQ_REGISTRY.register('tcp')
class TCPMQ(MQ):

    def __init__(self, listen_to: str, attach_to: Optional[List[str]] = None, **kwargs) -> None:
        self.listen_to = listen_to
        self.attach_to = attach_to or []
        self._recv_queue: "queue.Queue[Tuple[str, bytes]]" = queue.Queue()
        self._server: Optional[socket.socket] = None
        self._peers = {}
        self._peer_lock = threading.Lock()
        self._end = False
        self._threads: List[threading.Thread] = []

    # ---------------------------------------------------------- lifecycle
    def listen(self) -> None:
        host, port = _parse_addr(self.listen_to)
        self._server = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._server.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._server.bind((host, port))
        self._server.listen(64)
        t = threading.Thread(target=self._accept_loop, daemon=True)
        t.start()
        self._threads.append(t)
        for peer in self.attach_to:
            self._dial(peer)

    def _dial(self, addr: str, retries: int = 100, wait: float = 0.1):
        host, port = _parse_addr(addr)
        for _ in range(retries):
            try:
                s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
                s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                s.connect((host, port))
                with self._peer_lock:
                    self._peers[addr] = s
                # full duplex: peers publish back over the same link
                t = threading.Thread(target=self._recv_loop, args=(s, ), daemon=True)
                t.start()
                self._threads.append(t)
                return
            except (ConnectionRefusedError, OSError):
                time.sleep(wait)
        raise ConnectionError(f"cannot dial peer {addr}")

    def _accept_loop(self):
        n_accepted = 0
        while not self._end:
            try:
                conn, addr = self._server.accept()
            except OSError:
                break
            conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            # bus semantics: accepted links are full-duplex publish targets too
            with self._peer_lock:
                self._peers[f"accepted://{addr[0]}:{addr[1]}:{n_accepted}"] = conn
            n_accepted += 1
            t = threading.Thread(target=self._recv_loop, args=(conn, ), daemon=True)
            t.start()
            self._threads.append(t)

    def _recv_loop(self, conn: socket.socket):
        try:
            while not self._end:
                header = self._recv_exact(conn, _HEADER.size)
                if header is None:
                    break
                (length, ) = _HEADER.unpack(header)
                frame = self._recv_exact(conn, length)
                if frame is None:
                    break
                topic, _, payload = frame.partition(b"::")
                self._recv_queue.put((topic.decode(), payload))
        finally:
            try:
                conn.close()
            except OSError:
                pass

    @staticmethod
    def _recv_exact(conn: socket.socket, n: int) -> Optional[bytes]:
        buf = b""
        while len(buf) < n:
            try:
                chunk = conn.recv(n - len(buf))
            except OSError:
                return None
            if not chunk:
                return None
            buf += chunk
        return buf

    # ------------------------------------------------------------- pub/sub
    def publish(self, topic: str, data: bytes) -> None:
        frame = topic.encode() + b"::" + data
        msg = _HEADER.pack(len(frame)) + frame
        dead = []
        with self._peer_lock:
            peers = list(self._peers.items())
        for addr, s in peers:
            try:
                s.sendall(msg)
            except OSError:
                dead.append(addr)
        for addr in dead:
            with self._peer_lock:
                self._peers.pop(addr, None)

    def subscribe(self, topic: str) -> None:
        pass  # bus semantics: everything is delivered, filtering is local

    def unsubscribe(self, topic: str) -> None:
        pass

    def recv(self) -> Tuple[str, bytes]:
        while not self._end:
            try:
                return self._recv_queue.get(timeout=0.5)
            except queue.Empty:
                continue
        raise ConnectionAbortedError("mq stopped")

    def stop(self) -> None:
        self._end = True
        if self._server is not None:
            try:
                self._server.close()
            except OSError:
                pass
        with self._peer_lock:
            for s in self._peers.values():
                try:
                    s.close()
                except OSError:
                    pass
            self._peers = {}
