"""Distinct-ID monitoring: HyperLogLog estimator per feature.

Mirrors the reference embedding worker's HyperLogLog++ monitor
(rust/persia-embedding-server/src/monitor.rs:29-114): estimates the distinct
sign count each slot has produced, exported as a gauge (drives capacity
planning for the HBM table)."""
import math
import threading
from typing import Dict

import numpy as np

from persia_amd.core import hashing


class HyperLogLog:
    """Plain HLL (dense representation) over u64 hashes."""

    def __init__(self, p: int = 14):
        self.p = p
        self.m = 1 << p
        self.registers = np.zeros(self.m, dtype=np.uint8)
        if p == 14:
            self.alpha = 0.7213 / (1 + 1.079 / self.m)
        else:
            self.alpha = 0.7213 / (1 + 1.079 / self.m)

    def add_hashed(self, h: np.ndarray) -> None:
        """h: uint64 array of already-mixed hashes."""
        h = h.astype(np.uint64)
        idx = (h >> np.uint64(64 - self.p)).astype(np.int64)
        rest = h << np.uint64(self.p)
        # rank = leading zeros of the remaining 64-p bits + 1
        lz = np.full(len(h), 64 - self.p + 1, dtype=np.uint8)
        nonzero = rest != 0
        if nonzero.any():
            r = rest[nonzero]
            # count leading zeros via float exponent trick is lossy; do bitwise
            shift = np.zeros(r.shape, dtype=np.uint8)
            v = r.copy()
            for s in (32, 16, 8, 4, 2, 1):
                mask = v < (np.uint64(1) << np.uint64(64 - s))
                v[mask] = v[mask] << np.uint64(s)
                shift[mask] += s
            lz_nz = shift + 1
            lz[nonzero] = np.minimum(lz_nz, 64 - self.p + 1)
        np.maximum.at(self.registers, idx, lz)

    def add_signs(self, signs: np.ndarray) -> None:
        self.add_hashed(hashing.splitmix64(signs.astype(np.uint64)))

    def estimate(self) -> float:
        inv = np.power(2.0, -self.registers.astype(np.float64))
        e = self.alpha * self.m * self.m / inv.sum()
        zeros = int((self.registers == 0).sum())
        if e <= 2.5 * self.m and zeros > 0:
            e = self.m * math.log(self.m / zeros)  # small-range correction
        return e


class DistinctIdMonitor:
    """Per-slot HLLs fed from batch sign arrays (thread-safe)."""

    def __init__(self, p: int = 14):
        self.p = p
        self._hlls: Dict[str, HyperLogLog] = {}
        self._lock = threading.Lock()

    def observe(self, slot: str, signs: np.ndarray) -> None:
        with self._lock:
            hll = self._hlls.get(slot)
            if hll is None:
                hll = self._hlls[slot] = HyperLogLog(self.p)
        hll.add_signs(signs)

    def estimates(self) -> Dict[str, float]:
        with self._lock:
            return {k: v.estimate() for k, v in self._hlls.items()}
