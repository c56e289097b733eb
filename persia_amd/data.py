"""Data pipeline (mirrors reference persia/data.py).

* :class:`IterableDatasetBase` / :class:`StreamingDataset` /
  :class:`IterableDataset` — batch sources (data.py:83-199)
* :class:`DataLoader` — drives the engine's async lookup pipeline and yields
  device-ready :class:`PersiaTrainingBatch` (data.py:228-271; the native
  Forward engine's prefetch/staleness behavior is
  persia_amd.core.engine.ForwardPipeline)
"""
import threading
import time
from abc import ABC, abstractmethod
from typing import Iterator, Optional

from persia_amd.core.engine import ForwardPipeline, PersiaTrainingBatch
from persia_amd.core.queue import PersiaBatchDataChannel, get_default_channel
from persia_amd.ctx import cnt_ctx
from persia_amd.embedding.data import PersiaBatch
from persia_amd.logger import get_default_logger

_logger = get_default_logger("persia_amd.data")


class IterableDatasetBase(ABC):
    def __init__(self, buffer_size: int = 100):
        self.buffer_size = buffer_size

    @abstractmethod
    def batches(self) -> Iterator[PersiaBatch]:
        ...


class StreamingDataset(IterableDatasetBase):
    """Receives batches produced by a DataCtx (same process, or a separate
    loader process over the TCP queue — reference data.py:120-138)."""

    def __init__(self, buffer_size: int = 100, channel: Optional[PersiaBatchDataChannel] = None, port: Optional[int] = None):
        super().__init__(buffer_size)
        self.channel = channel or get_default_channel(buffer_size)
        self._server = None
        if port is not None:
            from persia_amd.core.queue import BatchQueueServer

            self._server = BatchQueueServer(port, self.channel)
        self._receiver = self.channel.get_receiver()

    def batches(self) -> Iterator[PersiaBatch]:
        while True:
            batch = self._receiver.recv(timeout=0.5)
            if batch is not None:
                yield batch


class IterableDataset(IterableDatasetBase):
    """Subclass and implement ``__iter__`` yielding PersiaBatch
    (reference data.py:141-199)."""

    def __iter__(self) -> Iterator[PersiaBatch]:
        raise NotImplementedError

    def batches(self) -> Iterator[PersiaBatch]:
        return iter(self)


class DataLoader:
    """reference data.py:202-271."""

    def __init__(
        self,
        dataset: IterableDatasetBase,
        forward_buffer_size: int = 10,
        timeout_ms: int = 1000 * 60 * 10,
        num_workers: int = 10,
        reproducible: bool = False,
        embedding_staleness: Optional[int] = None,
    ):
        self.dataset = dataset
        self.forward_buffer_size = forward_buffer_size
        self.timeout = timeout_ms / 1000.0
        self.num_workers = num_workers
        self.reproducible = reproducible
        self.embedding_staleness = embedding_staleness or 8
        self._pipeline: Optional[ForwardPipeline] = None
        self._last_slow_warn = 0.0

    def _ensure_pipeline(self) -> ForwardPipeline:
        if self._pipeline is None:
            ctx = cnt_ctx()
            assert ctx is not None and hasattr(ctx, "engine"), (
                "DataLoader must be iterated inside an EmbeddingCtx/TrainCtx"
            )
            staleness = 1 if self.reproducible else self.embedding_staleness
            self._pipeline = ForwardPipeline(
                ctx.engine, staleness=staleness, out_buffer=self.forward_buffer_size,
                reorder=self.reproducible,
            )
            if hasattr(ctx, "_pipeline"):
                ctx._pipeline = self._pipeline
        return self._pipeline

    def __iter__(self) -> Iterator[PersiaTrainingBatch]:
        pipeline = self._ensure_pipeline()
        done = threading.Event()

        def feed():
            try:
                for batch in self.dataset.batches():
                    if done.is_set():
                        return
                    pipeline.put(batch)
            finally:
                pipeline.finish()

        feeder = threading.Thread(target=feed, daemon=True, name="persia-data-feeder")
        feeder.start()
        try:
            while True:
                _t0 = time.perf_counter()
                tb = pipeline.get(timeout=self.timeout)
                _wait = time.perf_counter() - _t0
                if _wait > 1e-3 and _t0 - self._last_slow_warn > 10.0:
                    # reference forward.rs:860-897 warns when get_batch
                    # blocks >1 ms (trainer starving on the lookup
                    # pipeline); rate-limited to one warning per 10 s
                    self._last_slow_warn = _t0
                    _logger.warning(
                        f"get_batch waited {_wait*1e3:.1f} ms; consider a "
                        f"larger embedding_staleness/forward_buffer_size"
                    )
                if tb is None:
                    break
                if not tb.requires_grad:
                    # no backward will come; slide the staleness window now
                    pipeline.release_permit()
                yield tb
        finally:
            done.set()
            pipeline.stop()
            self._pipeline = None
