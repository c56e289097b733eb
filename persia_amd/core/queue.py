"""Batch dataflow channels.

The reference moves whole batches loader→trainer over NATS
(persia-core/src/nats.rs:102-143 DataflowService) and exposes a native
bounded channel (persia-core/src/utils.rs PersiaBatchDataChannel).  On one
node this collapses to an in-process bounded queue; a TCP variant covers
separate data-loader processes (reference PersiaMessageQueue analog).
"""
import queue
import socket
import struct
import threading
from typing import Optional

from persia_amd.embedding.data import PersiaBatch
from persia_amd.logger import get_default_logger

_logger = get_default_logger("persia_amd.queue")


class PersiaBatchDataChannel:
    """Bounded in-process channel (reference persia-core/src/utils.rs)."""

    def __init__(self, buffer_size: int = 100):
        self._q: "queue.Queue" = queue.Queue(maxsize=buffer_size)

    def get_sender(self) -> "PersiaBatchDataSender":
        return PersiaBatchDataSender(self)

    def get_receiver(self) -> "PersiaBatchDataReceiver":
        return PersiaBatchDataReceiver(self)


class PersiaBatchDataSender:
    def __init__(self, channel: PersiaBatchDataChannel):
        self._channel = channel

    def send(self, batch: PersiaBatch, block: bool = True):
        self._channel._q.put(batch, block=block)


class PersiaBatchDataReceiver:
    def __init__(self, channel: PersiaBatchDataChannel):
        self._channel = channel

    def recv(self, timeout: Optional[float] = None) -> Optional[PersiaBatch]:
        try:
            return self._channel._q.get(timeout=timeout)
        except queue.Empty:
            return None


_default_channel: Optional[PersiaBatchDataChannel] = None
_lock = threading.Lock()


def get_default_channel(buffer_size: int = 100) -> PersiaBatchDataChannel:
    global _default_channel
    with _lock:
        if _default_channel is None:
            _default_channel = PersiaBatchDataChannel(buffer_size)
        return _default_channel


def get_default_sink() -> PersiaBatchDataSender:
    return get_default_channel().get_sender()


# --------------------------------------------------------------- TCP variant


class BatchQueueServer:
    """Listens for length-prefixed PersiaBatch bytes and feeds a channel
    (trainer side of a separate data-loader process)."""

    def __init__(self, port: int, channel: Optional[PersiaBatchDataChannel] = None, host: str = "127.0.0.1"):
        self.channel = channel or get_default_channel()
        self._sender = self.channel.get_sender()
        self._sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._sock.bind((host, port))
        self._sock.listen(16)
        self.port = self._sock.getsockname()[1]
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._accept_loop, daemon=True)
        self._thread.start()

    def _accept_loop(self):
        while not self._stop.is_set():
            try:
                self._sock.settimeout(0.5)
                conn, _ = self._sock.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            threading.Thread(target=self._recv_loop, args=(conn,), daemon=True).start()

    def _recv_loop(self, conn: socket.socket):
        try:
            while not self._stop.is_set():
                hdr = _recv_exact(conn, 8)
                if hdr is None:
                    break
                (n,) = struct.unpack("<Q", hdr)
                payload = _recv_exact(conn, n)
                if payload is None:
                    break
                self._sender.send(PersiaBatch.from_bytes(payload))
        finally:
            conn.close()

    def close(self):
        self._stop.set()
        try:
            self._sock.close()
        except OSError:
            pass


class BatchQueueClient:
    """Data-loader side: pushes serialized batches to a trainer."""

    def __init__(self, host: str, port: int, timeout: float = 30.0):
        self._sock = socket.create_connection((host, port), timeout=timeout)

    def send(self, batch: PersiaBatch, block: bool = True):
        payload = batch.to_bytes()
        self._sock.sendall(struct.pack("<Q", len(payload)) + payload)

    def close(self):
        self._sock.close()


def _recv_exact(conn: socket.socket, n: int) -> Optional[bytes]:
    buf = b""
    while len(buf) < n:
        chunk = conn.recv(n - len(buf))
        if not chunk:
            return None
        buf += chunk
    return buf
