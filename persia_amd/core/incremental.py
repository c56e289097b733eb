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
    """Train-side: collect touched signs, flush packets."""

    def __init__(self, engine: "EmbeddingEngine", dst_dir: str,
                 buffer_size: int = 1_000_000):
        self.engine = engine
        self.dst_dir = dst_dir
        self.buffer_size = buffer_size
        self._touched: Set[int] = set()
        self._lock = threading.Lock()
        self._seq = 0

    def record(self, signs: np.ndarray) -> None:
        flush_signs = None
        with self._lock:
            self._touched.update(int(s) for s in signs)
            if len(self._touched) >= self.buffer_size:
                flush_signs = self._take()
        if flush_signs is not None:
            self._dump(flush_signs)

    def _take(self) -> Optional[np.ndarray]:
        if not self._touched:
            return None
        signs = np.fromiter(self._touched, dtype=np.uint64, count=len(self._touched))
        self._touched.clear()
        return signs

    def flush(self) -> Optional[str]:
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
        # export current rows for the touched signs per dim store
        keys = hashing.splitmix64(signs)
        for i, dim in enumerate(sorted(self.engine.stores.keys())):
            store = self.engine.stores[dim]
            import torch

            rows = store.lookup(
                torch.from_numpy(keys.view(np.int64)).to(self.engine.device),
                train=False,
            )
            present = ~(rows == 0).all(dim=1)
            present_np = present.cpu().numpy()
            if present_np.any():
                # re-export full rows (emb + opt state) for resident signs
                all_signs, all_inner = store.export_rows()
                keep = np.isin(all_signs, signs[present_np])
                write_emb_file(
                    os.path.join(pkt_dir, f"{rank}_{i}.inc"),
                    all_signs[keep], all_inner[keep], dim,
                )
        with open(os.path.join(pkt_dir, DONE_MARKER), "w", encoding="utf-8") as f:
            f.write(str(ts))
        self._seq += 1
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
                signs, inner, dim = read_emb_file(os.path.join(full, fn))
                if dim in self.engine.stores and len(signs):
                    self.engine.stores[dim].import_rows(signs, inner)
                    loaded += len(signs)
            self._seen.add(name)
            try:
                ts = int(name.split("_", 1)[1]) / 1000.0
                self.last_delay_sec = max(0.0, time.time() - ts)
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
