"""Test/dev harness (mirrors reference persia/helper.py PersiaServiceCtx).

The reference spawns nats-server + Rust worker/parameter-server binaries as
subprocesses; here the embedding tier is in-process, so the harness only has
to provide: a fresh default dataflow channel, an engine-backed ctx factory
and optional data-loader threads."""
import threading
from typing import Callable, List, Optional

from persia_amd.core import queue as _q
from persia_amd.core.queue import PersiaBatchDataChannel
from persia_amd.logger import get_default_logger

_logger = get_default_logger("persia_amd.helper")


class PersiaServiceCtx:
    """Single-node mock "cluster": isolates the default dataflow channel and
    runs data-loader functions on threads (reference helper.py:125-331)."""

    def __init__(
        self,
        data_loader_func: Optional[Callable] = None,
        num_data_loaders: int = 1,
        buffer_size: int = 100,
    ):
        self.data_loader_func = data_loader_func
        self.num_data_loaders = num_data_loaders
        self.buffer_size = buffer_size
        self._threads: List[threading.Thread] = []
        self._prev_channel = None

    def __enter__(self):
        self._prev_channel = _q._default_channel
        _q._default_channel = PersiaBatchDataChannel(self.buffer_size)
        if self.data_loader_func is not None:
            for i in range(self.num_data_loaders):
                t = threading.Thread(
                    target=self.data_loader_func, daemon=True, name=f"persia-loader-{i}"
                )
                t.start()
                self._threads.append(t)
        return self

    def __exit__(self, exc_type, value, trace):
        _q._default_channel = self._prev_channel
        return False
