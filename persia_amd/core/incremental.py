"""Incremental-update manager: online-inference freshness.

Mirrors the reference ``persia-incremental-update-manager`` crate
(persia-incremental-update-manager/src/lib.rs:46-364):

* train side: buffer the signs touched by gradient updates (dedup set) until
  ``buffer_size``, then dump a packet dir ``inc_{timestamp}/{rank}_{i}.inc``
  (same record format as checkpoints) + ``inc_update_done`` marker;
* infer side: scan the base dir periodically, load packets newer than the
  last seen, export an ``inc_update_delay_sec`` gauge.
"""
import os
import threading
import time
from typing import TYPE_CHECKING, Optional, Set

import numpy as np

from persia_amd.core import hashing
from persia_amd.core.checkpoint import read_emb_file, write_emb_file
from persia_amd.logger import get_default_logger

if TYPE_CHECKING:
    from persia_amd.core.engine import EmbeddingEngine

_logger = get_default_logger("persia_amd.incremental")

DONE_MARKER = "inc_update_done"


class IncrementalUpdateDumper:
    """Train-side: collect the signs touched by gradient updates, flush
    packets automatically when the dedup buffer fills (reference
    lib.rs:178-312 buffers inside update_gradient the same way).

    The engine calls :meth:`record_keys` from its update paths with the
    DEVICE key tensor of the batch; the copy back to the host is asynchronous
    (pinned staging + event, harvested on later calls), so recording costs
    the hot loop one small async D2H, never a stream sync."""

    def __init__(self, engine: "EmbeddingEngine", dst_dir: str,
                 buffer_size: int = 1_000_000):
        self.engine = engine
        self.dst_dir = dst_dir
        self.buffer_size = buffer_size
        self._touched: Set[int] = set()
        self._lock = threading.Lock()
        self._seq = 0
        self._pending: list = []  # [(pinned keys, event or None)]

    # ------------------------------------------------------------- recording

    def record(self, signs: np.ndarray) -> None:
        """Record touched SIGNS (u64 id space) directly."""
        flush_signs = None
        with self._lock:
            self._touched.update(int(s) for s in signs)
            if len(self._touched) >= self.buffer_size:
                flush_signs = self._take()
        if flush_signs is not None:
            self._dump(flush_signs)

    def record_keys(self, keys, u_count=None) -> None:
        """Record touched MIXED keys from a device (or CPU) int64 tensor.
        GPU tensors are staged through a pinned async copy; ``u_count`` is the
        device-side valid-prefix length of a padded dedup (the padding tail is
        key 0 and is dropped at harvest)."""
        import torch

        if keys.is_cuda:
            n = keys.numel()
            pin = torch.empty(n + 1, dtype=torch.int64, pin_memory=True)
            pin[:n].copy_(keys, non_blocking=True)
            if u_count is not None:
                pin[n:].copy_(u_count, non_blocking=True)
            else:
                pin[n] = n
            ev = torch.cuda.Event()
            ev.record()
            with self._lock:
                self._pending.append((pin, ev))
            self._harvest(block=False)
        else:
            k_np = keys.numpy().view(np.uint64)
            self.record(hashing.splitmix64_inv(k_np[k_np != 0]))

    def _harvest(self, block: bool) -> None:
        """Convert completed pending copies into signs."""
        with self._lock:
            ready, still = [], []
            for pin, ev in self._pending:
                if block:
                    ev.synchronize()
                if ev.query():
                    ready.append(pin)
                else:
                    still.append((pin, ev))
            self._pending = still
        for pin in ready:
            n_valid = int(pin[-1].item())
            k_np = pin[:n_valid].numpy().view(np.uint64)
            self.record(hashing.splitmix64_inv(k_np[k_np != 0]))

    def _take(self) -> Optional[np.ndarray]:
        if not self._touched:
            return None
        signs = np.fromiter(self._touched, dtype=np.uint64, count=len(self._touched))
        self._touched.clear()
        return signs

    def flush(self) -> Optional[str]:
        self._harvest(block=True)
        with self._lock:
            signs = self._take()
        if signs is None:
            return None
        return self._dump(signs)

    def _dump(self, signs: np.ndarray) -> str:
        ts = int(time.time() * 1000)
        pkt_dir = os.path.join(self.dst_dir, f"inc_{ts}")
        os.makedirs(pkt_dir, exist_ok=True)
        rank = self.engine.dist.rank
        for i, dim in enumerate(sorted(self.engine.stores.keys())):
            store = self.engine.stores[dim]
            # probe-based presence + row export for exactly the touched signs
            # (store.export_keys; never a full-table export, no insert, and a
            # legitimately all-zero row is still exported)
            found_signs, inner = store.export_keys(signs)
            if len(found_signs):
                write_emb_file(
                    os.path.join(pkt_dir, f"{rank}_{i}.inc"),
                    found_signs, inner, dim,
                )
        with open(os.path.join(pkt_dir, DONE_MARKER), "w", encoding="utf-8") as f:
            f.write(str(ts))
        self._seq += 1
        if self.engine.metrics_enabled:
            self.engine.metrics.inc_packets_dumped.inc(1)
        return pkt_dir


class IncrementalUpdateLoader:
    """Infer-side: poll for new packets and apply them."""

    def __init__(self, engine: "EmbeddingEngine", src_dir: str,
                 poll_interval_sec: float = 10.0):
        self.engine = engine
        self.src_dir = src_dir
        self.poll_interval = poll_interval_sec
        self._seen: Set[str] = set()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.last_delay_sec: float = 0.0

    def scan_once(self) -> int:
        if not os.path.isdir(self.src_dir):
            return 0
        loaded = 0
        for name in sorted(os.listdir(self.src_dir)):
            full = os.path.join(self.src_dir, name)
            if not name.startswith("inc_") or name in self._seen:
                continue
            if not os.path.exists(os.path.join(full, DONE_MARKER)):
                continue
            for fn in sorted(os.listdir(full)):
                if not fn.endswith(".inc"):
                    continue
                try:
                    signs, inner, dim = read_emb_file(os.path.join(full, fn))
                except Exception as e:
                    # a torn/corrupt packet file must not wedge freshness:
                    # skip it (the packet dir is marked seen either way)
                    _logger.warning(f"skipping corrupt packet {full}/{fn}: {e}")
                    continue
                if dim in self.engine.stores and len(signs):
                    self.engine.stores[dim].import_rows(signs, inner)
                    loaded += len(signs)
            self._seen.add(name)
            try:
                ts = int(name.split("_", 1)[1]) / 1000.0
                self.last_delay_sec = max(0.0, time.time() - ts)
                if self.engine.metrics_enabled:
                    # reference inc-update lib.rs:48-65 gauge
                    self.engine.metrics.inc_update_delay_sec.set(
                        self.last_delay_sec
                    )
            except ValueError:
                pass
        return loaded

    def start(self):
        def loop():
            while not self._stop.is_set():
                try:
                    self.scan_once()
                except Exception as e:
                    _logger.warning(f"incremental scan failed: {e}")
                self._stop.wait(self.poll_interval)

        self._thread = threading.Thread(target=loop, daemon=True, name="persia-inc-loader")
        self._thread.start()

    def stop(self):
        self._stop.set()
