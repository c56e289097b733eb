"""The embedding store: one shard of the (conceptually trillion-row) table.

Re-architecture of the reference parameter-server storage stack
(`persia-embedding-holder`: Sharded<EvictionMap<u64, HashMapEmbeddingEntry>>,
LRU via ArrayLinkedList — rust/persia-embedding-holder/src/lib.rs:28-64) as a
GPU-friendly **set-associative hash table**:

* keys:  ``uint64[n_slots]``   — splitmix64-mixed signs (bijective, so the
  mixed key IS the hash; 0 = empty sentinel)
* ticks: ``uint32[n_slots]``   — last-access batch counter (approximate LRU:
  evict the min-tick slot of the probe window when full)
* arena: ``float32[n_slots, row_width]`` — row = [emb(dim) | opt_state]
  (same logical layout as the reference's HashMapEmbeddingEntry.inner)

Buckets of ``BUCKET_SIZE`` slots; probing scans ``PROBE_BUCKETS`` consecutive
buckets.  Eviction is bounded-window LRU instead of the reference's global
linked-list LRU — O(1), no global state, GPU-parallel.

Two implementations with identical semantics:
* :class:`CpuEmbeddingStore` — numpy/torch sequential model (the oracle;
  also the CPU execution backend).
* :class:`HipEmbeddingStore` — HBM-resident, HIP kernels via persia_amd._C.
"""
from typing import Optional, Tuple

import numpy as np
import torch

from persia_amd.core import hashing
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.optim import Adagrad, Adam, Optimizer, SGD

BUCKET_SIZE = 8
PROBE_BUCKETS = 4
EMPTY_KEY = np.uint64(0)
_ZERO_REMAP = np.uint64(0xD1B54A32D192ED03)  # mixed-key 0 is remapped here


def _floor_pow2(x: int) -> int:
    """Largest power of two <= x (capacity is a BUDGET: never round up —
    a ceil on a 2.6e9-row request would double a 128 GB arena)."""
    p = 1
    while (p << 1) <= x:
        p <<= 1
    return p


def _u01_from_u64(u: np.ndarray) -> np.ndarray:
    """Uniform [0,1) from the top 24 bits — must match csrc/common.h."""
    return (u >> np.uint64(40)).astype(np.float64) * (1.0 / (1 << 24))


def row_init(sign: int, dim: int, lo: float, hi: float) -> np.ndarray:
    """Deterministic per-sign bounded-uniform init (reference seeds SmallRng
    by sign — emb_entry.rs:35; we use splitmix64 streams, same both CPU/HIP)."""
    seed = hashing.init_seed_for(np.array([sign], dtype=np.uint64))[0]
    cols = np.arange(1, dim + 1, dtype=np.uint64)
    u = hashing.splitmix64(np.uint64(seed) ^ cols)
    return (lo + (hi - lo) * _u01_from_u64(u)).astype(np.float32)


class HostTier:
    """Host-DRAM spill tier: rows evicted from the HBM table park here and
    come back on their next lookup (exclusive cache: a fetch removes the
    entry, an eviction re-inserts the freshest copy).  Bounded LRU.

    Replaces the reference's remote-CPU parameter servers as the capacity
    extension beyond HBM (BASELINE config 4: 1e11-row tables)."""

    def __init__(self, capacity: int, row_width: int):
        from collections import OrderedDict

        self.capacity = capacity
        self.row_width = row_width
        self._map: "OrderedDict[int, np.ndarray]" = OrderedDict()

    def __len__(self) -> int:
        return len(self._map)

    def insert(self, keys: np.ndarray, rows: np.ndarray) -> None:
        for i, k in enumerate(keys):
            k = int(k)
            if k in self._map:
                self._map.move_to_end(k)
            self._map[k] = rows[i].copy()
        while len(self._map) > self.capacity:
            self._map.popitem(last=False)

    def fetch(self, keys: np.ndarray):
        """-> (rows f32 [k, row_width], found bool [k]); found entries are
        removed (they move to the HBM tier)."""
        rows = np.zeros((len(keys), self.row_width), dtype=np.float32)
        found = np.zeros(len(keys), dtype=bool)
        for i, k in enumerate(keys):
            row = self._map.pop(int(k), None)
            if row is not None:
                rows[i] = row
                found[i] = True
        return rows, found

    def __contains__(self, key) -> bool:
        return int(key) in self._map

    def export(self):
        """-> (keys u64 [n], rows f32 [n, row_width]) for checkpointing."""
        n = len(self._map)
        keys = np.fromiter(self._map.keys(), dtype=np.uint64, count=n)
        rows = (
            np.stack(list(self._map.values()))
            if n
            else np.zeros((0, self.row_width), dtype=np.float32)
        )
        return keys, rows.astype(np.float32)


class NativeHostTier:
    """C++ host spill tier (csrc/engine.cpp NativeHostTier) behind the same
    numpy interface as :class:`HostTier` — the python dict loops were the
    dominant producer-thread cost of spill-heavy configs."""

    def __init__(self, capacity: int, row_width: int):
        from persia_amd.ops import native

        self.capacity = capacity
        self.row_width = row_width
        self._impl = native().HostTier(capacity, row_width)

    def __len__(self) -> int:
        return int(self._impl.size())

    def insert(self, keys: np.ndarray, rows: np.ndarray) -> None:
        self._impl.insert(
            torch.from_numpy(np.ascontiguousarray(keys).view(np.int64)),
            torch.from_numpy(np.ascontiguousarray(rows, dtype=np.float32)),
        )

    def fetch(self, keys: np.ndarray):
        rows, found = self._impl.fetch(
            torch.from_numpy(np.ascontiguousarray(keys).view(np.int64))
        )
        return rows.numpy(), found.numpy()

    def __contains__(self, key) -> bool:
        return bool(self._impl.contains(int(np.int64(np.uint64(key)))))

    def export(self):
        keys, rows = self._impl.export_all()
        return keys.numpy().view(np.uint64), rows.numpy()



def _adapt_row_width(inner: np.ndarray, row_width: int,
                     state_init: float = 0.0) -> np.ndarray:
    """Adapt checkpoint rows to this store's row width: an infer-role store
    (no optimizer state) truncates the dumped opt-state columns; a train
    store loading an optimizer-less dump re-initializes the state columns
    (reference infer PS keeps whatever inner was dumped and slices the emb
    prefix on lookup — mod.rs:231-251; adapting at import keeps our fixed
    arena layout)."""
    cur = inner.shape[1]
    if cur == row_width:
        return inner
    if cur > row_width:
        return np.ascontiguousarray(inner[:, :row_width])
    out = np.full((inner.shape[0], row_width), state_init, dtype=np.float32)
    out[:, :cur] = inner
    return out


class EmbeddingStoreBase:
    """One rank's shard for one dim-group of slots."""

    def __init__(
        self,
        dim: int,
        capacity: int,
        optimizer: Optimizer,
        hyper: EmbeddingConfig,
        device: torch.device,
        spill_capacity: int = 0,
    ):
        self.dim = dim
        self.optimizer = optimizer
        self.hyper = hyper
        self.device = device
        self.opt_space = optimizer.require_space(dim)
        self.row_width = dim + self.opt_space
        self.n_buckets = _floor_pow2(max(1, capacity // BUCKET_SIZE))
        self.n_slots = self.n_buckets * BUCKET_SIZE
        self.tick = 1  # current batch counter (0 reserved)
        # tick allocation may be hit from the pipeline thread (lookups) and
        # the main thread (updates, incremental exports) concurrently
        self._tick_lock = __import__("threading").Lock()
        self.spill = HostTier(spill_capacity, self.row_width) if spill_capacity > 0 else None
        # Adam per-group beta powers (persia-common optim.rs:147-216); kept
        # per-store and stepped once per update call (all slot-groups in a
        # store receive every update batch in this architecture).
        if isinstance(optimizer, Adam):
            self.beta1_power = optimizer.betas[0]
            self.beta2_power = optimizer.betas[1]

    def __len__(self) -> int:
        raise NotImplementedError

    def lookup(self, keys: torch.Tensor, train: bool,
               u_count: Optional[torch.Tensor] = None) -> torch.Tensor:
        """keys: int64[n] (mixed u64 bit pattern) -> rows float32[n, dim].
        train=True inserts on miss (admit-gated, seeded init);
        train=False returns zeros on miss (reference parameter
        mod.rs:231-251).  ``u_count``: HIP-only padded-prefix length."""
        raise NotImplementedError

    def update_gradients(self, keys: torch.Tensor, grads: torch.Tensor) -> int:
        """Apply the sparse optimizer to rows addressed by keys; skip keys no
        longer resident. Returns number skipped."""
        raise NotImplementedError

    def lookup_wire(self, keys: torch.Tensor, train: bool,
                    wire_dtype: torch.dtype) -> torch.Tensor:
        """Owner-side lookup for the padded a2a exchange: ``keys`` may contain
        the empty-key sentinel 0 (bucket padding) — those entries are silently
        skipped (zeros out, no insert).  Rows come back in the WIRE dtype so
        they go straight onto the xGMI all-to-all."""
        raise NotImplementedError

    def export_rows(self) -> Tuple[np.ndarray, np.ndarray]:
        """-> (signs u64[n], inner f32[n, row_width]) of every resident row."""
        raise NotImplementedError

    def import_rows(self, signs: np.ndarray, inner: np.ndarray) -> None:
        raise NotImplementedError

    def clear(self) -> None:
        raise NotImplementedError

    def next_tick(self) -> int:
        with self._tick_lock:
            t = self.tick
            self.tick += 1
        return t

    def export_keys(self, signs: np.ndarray) -> Tuple[np.ndarray, np.ndarray]:
        """Export rows for specific signs: (found_signs, inner[found]).
        Presence comes from the table probe (NOT a rows==0 heuristic — an
        all-zero row is a legitimate value), without insert-on-miss and
        without consuming a tick.  Used by the incremental-update dumper;
        may run concurrently with training — a row mutated mid-read yields a
        mixed old/new row, which the bounded-staleness freshness contract
        tolerates (the reference holds per-sign locks instead)."""
        raise NotImplementedError

    def _spill_export(self):
        """Host-tier rows for checkpointing (spilled rows are table state)."""
        if self.spill is None or len(self.spill) == 0:
            return None
        keys, rows = self.spill.export()
        return hashing.splitmix64_inv(keys), rows

    def _adam_step_powers(self):
        o = self.optimizer
        if isinstance(o, Adam):
            b1p, b2p = self.beta1_power, self.beta2_power
            self.beta1_power *= o.betas[0]
            self.beta2_power *= o.betas[1]
            return b1p, b2p
        return None


class CpuEmbeddingStore(EmbeddingStoreBase):
    """Sequential oracle + CPU execution backend."""

    def __init__(self, dim, capacity, optimizer, hyper, device=torch.device("cpu"),
                 spill_capacity: int = 0):
        super().__init__(dim, capacity, optimizer, hyper, device, spill_capacity)
        self.keys = np.zeros(self.n_slots, dtype=np.uint64)
        self.ticks = np.zeros(self.n_slots, dtype=np.uint32)
        self.arena = torch.zeros(self.n_slots, self.row_width, dtype=torch.float32)
        self._count = 0

    def __len__(self) -> int:
        return self._count

    # -- probing ------------------------------------------------------------
    def _probe(self, k: np.uint64) -> int:
        """Find slot of key k, or -1."""
        mask = np.uint64(self.n_buckets - 1)
        b0 = int(k & mask)
        for p in range(PROBE_BUCKETS):
            base = ((b0 + p) % self.n_buckets) * BUCKET_SIZE
            for s in range(BUCKET_SIZE):
                if self.keys[base + s] == k:
                    return base + s
        return -1

    def _probe_or_claim(self, k: np.uint64, tick: int) -> Tuple[int, bool]:
        """-> (slot, is_new). Claims an empty slot, else evicts the min-tick
        slot of the probe window; rows touched at the CURRENT tick are never
        victims (same rule as the HIP kernel, so concurrent claims within a
        batch cannot evict each other). slot=-1 = overflow miss."""
        mask = np.uint64(self.n_buckets - 1)
        b0 = int(k & mask)
        empty = -1
        victim, victim_tick = -1, None
        for p in range(PROBE_BUCKETS):
            base = ((b0 + p) % self.n_buckets) * BUCKET_SIZE
            for s in range(BUCKET_SIZE):
                i = base + s
                ki = self.keys[i]
                if ki == k:
                    return i, False
                if ki == EMPTY_KEY:
                    if empty < 0:
                        empty = i
                elif self.ticks[i] != tick and (
                    victim_tick is None or self.ticks[i] < victim_tick
                ):
                    victim, victim_tick = i, self.ticks[i]
        if empty >= 0:
            self.keys[empty] = k
            self.ticks[empty] = tick
            self._count += 1
            return empty, True
        if victim < 0:
            return -1, False  # whole window is current-tick: overflow
        if self.spill is not None:  # save the victim row to the host tier
            self.spill.insert(
                np.array([self.keys[victim]], dtype=np.uint64),
                self.arena[victim].numpy().reshape(1, -1),
            )
        self.keys[victim] = k  # evict (bounded-window LRU)
        self.ticks[victim] = tick
        return victim, True

    def _init_row(self, slot: int, k: np.uint64) -> None:
        sign = int(hashing.splitmix64_inv(np.array([k], dtype=np.uint64))[0])
        lo, hi = self.hyper.emb_initialization
        emb = row_init(sign, self.dim, lo, hi)
        row = torch.zeros(self.row_width, dtype=torch.float32)
        row[: self.dim] = torch.from_numpy(emb)
        if self.opt_space:
            row[self.dim :] = self.optimizer.state_init(self.dim)
        self.arena[slot] = row

    def _admitted(self, k: np.uint64, tick: int) -> bool:
        if self.hyper.admit_probability >= 1.0:
            return True
        u = hashing.splitmix64(np.array([k ^ np.uint64(tick)], dtype=np.uint64))
        return float(_u01_from_u64(u)[0]) < self.hyper.admit_probability

    # -- public ops ---------------------------------------------------------
    def lookup(self, keys: torch.Tensor, train: bool,
               u_count: Optional[torch.Tensor] = None) -> torch.Tensor:
        assert u_count is None, "padded lookup is a HIP-path feature"
        tick = self.next_tick()
        k_np = keys.cpu().numpy().view(np.uint64).copy()
        k_np[k_np == EMPTY_KEY] = _ZERO_REMAP
        out = torch.zeros(len(k_np), self.dim, dtype=torch.float32)
        for i, k in enumerate(k_np):
            if train:
                if self._admitted(k, tick) or self._probe(k) >= 0:
                    slot, is_new = self._probe_or_claim(k, tick)
                    if slot < 0:
                        continue  # overflow miss: zeros
                    if is_new:
                        restored = False
                        if self.spill is not None:
                            rows, found = self.spill.fetch(np.array([k], dtype=np.uint64))
                            if found[0]:
                                self.arena[slot] = torch.from_numpy(rows[0])
                                restored = True
                        if not restored:
                            self._init_row(slot, k)
                    self.ticks[slot] = tick
                    out[i] = self.arena[slot, : self.dim]
            else:
                slot = self._probe(k)
                if slot >= 0:
                    self.ticks[slot] = tick
                    out[i] = self.arena[slot, : self.dim]
        return out

    def lookup_wire(self, keys: torch.Tensor, train: bool,
                    wire_dtype: torch.dtype) -> torch.Tensor:
        out = torch.zeros(keys.numel(), self.dim, dtype=torch.float32)
        m = keys != 0
        if bool(m.any()):
            out[m] = self.lookup(keys[m], train)
        else:
            self.next_tick()  # tick parity with the non-empty case
        return out.to(wire_dtype)

    def update_gradients(self, keys: torch.Tensor, grads: torch.Tensor) -> int:
        powers = self._adam_step_powers()
        k_np = keys.cpu().numpy().view(np.uint64).copy()
        k_np[k_np == EMPTY_KEY] = _ZERO_REMAP
        slots = np.array([self._probe(k) for k in k_np], dtype=np.int64)
        present = slots >= 0
        # row-level NaN skip (same semantics as the HIP update kernel)
        nan_rows = torch.isnan(grads.float()).any(dim=1).cpu().numpy()
        present = present & ~nan_rows
        skipped = int((slots < 0).sum())
        if present.sum() == 0:
            return skipped
        idx = torch.from_numpy(slots[present])
        g = grads.float().cpu()[torch.from_numpy(np.nonzero(present)[0])]
        rows = self.arena[idx]  # copy [n, row_width]
        emb = rows[:, : self.dim]
        from persia_amd.ops import reference as R

        o = self.optimizer
        wb = self.hyper.weight_bound
        if isinstance(o, SGD):
            R.sgd_update(emb, g, o.lr, o.weight_decay, wb)
        elif isinstance(o, Adagrad):
            accum = rows[:, self.dim :] if not o.vectorwise_shared else rows[:, self.dim : self.dim + 1]
            R.adagrad_update(
                emb, accum, g, o.lr, o.g_square_momentum, o.eps, wb, o.vectorwise_shared
            )
        elif isinstance(o, Adam):
            m = rows[:, self.dim : 2 * self.dim]
            v = rows[:, 2 * self.dim :]
            b1p, b2p = powers
            R.adam_update(emb, m, v, g, b1p, b2p, o.lr, o.betas[0], o.betas[1], o.eps, wb)
        else:
            raise ValueError(f"unknown optimizer {o.kind}")
        self.arena[idx] = rows
        return skipped

    def export_keys(self, signs: np.ndarray) -> Tuple[np.ndarray, np.ndarray]:
        ks = hashing.splitmix64(signs.astype(np.uint64))
        ks[ks == EMPTY_KEY] = _ZERO_REMAP
        slots = np.array([self._probe(k) for k in ks], dtype=np.int64)
        found = slots >= 0
        inner = self.arena[torch.from_numpy(slots[found])].numpy().copy()
        return signs[found], inner

    def export_rows(self) -> Tuple[np.ndarray, np.ndarray]:
        occ = np.nonzero(self.keys != EMPTY_KEY)[0]
        signs = hashing.splitmix64_inv(self.keys[occ])
        inner = self.arena[torch.from_numpy(occ)].numpy()
        sp = self._spill_export()
        if sp is not None:
            signs = np.concatenate([signs, sp[0]])
            inner = np.concatenate([inner, sp[1]])
        return signs, inner

    def import_rows(self, signs: np.ndarray, inner: np.ndarray) -> None:
        inner = _adapt_row_width(
            inner, self.row_width, float(self.optimizer.state_init(self.dim))
        )
        ks = hashing.splitmix64(signs.astype(np.uint64))
        ks[ks == EMPTY_KEY] = _ZERO_REMAP
        tick = self.next_tick()
        for i, k in enumerate(ks):
            slot, _ = self._probe_or_claim(k, tick)
            if slot < 0:
                continue  # shard over capacity: drop (bounded table)
            self.arena[slot] = torch.from_numpy(inner[i].astype(np.float32))
            self.ticks[slot] = tick

    def clear(self) -> None:
        self.keys[:] = EMPTY_KEY
        self.ticks[:] = 0
        self.arena.zero_()
        self._count = 0


class HipEmbeddingStore(EmbeddingStoreBase):
    """HBM-resident shard driven by HIP kernels (persia_amd._C).

    All state lives in torch CUDA tensors (allocated through the caching
    allocator so it coexists with the dense model); kernels mutate them in
    place."""

    def __init__(self, dim, capacity, optimizer, hyper, device, spill_capacity: int = 0):
        super().__init__(dim, capacity, optimizer, hyper, device, spill_capacity)
        from persia_amd.ops import native

        self._C = native()
        assert device.type == "cuda"
        if self.spill is not None:
            # native spill tier: same LRU semantics, no python dict loops
            self.spill = NativeHostTier(spill_capacity, self.row_width)
        self.keys = torch.zeros(self.n_slots, dtype=torch.int64, device=device)
        self.ticks = torch.zeros(self.n_slots, dtype=torch.int32, device=device)
        self.arena = torch.zeros(
            self.n_slots, self.row_width, dtype=torch.float32, device=device
        )
        self._skipped = torch.zeros(2, dtype=torch.int32, device=device)  # miss, nan
        self._opt_code = {"sgd": 0, "adagrad": 1, "adam": 2}[optimizer.kind]
        self._probe_stream: Optional[torch.cuda.Stream] = None
        self._restore_tick: Optional[int] = None
        self.evicted_total = 0  # host-side eviction count (spill drains)
        self._no_evict = (
            torch.empty(0, dtype=torch.int64, device=device),
            torch.empty(0, dtype=torch.int32, device=device),
            torch.empty(0, dtype=torch.float32, device=device),
        )

    def _evict_buffers(self, n: int):
        if self.spill is None:
            return self._no_evict
        return (
            torch.zeros(n, dtype=torch.int64, device=self.device),
            torch.zeros(1, dtype=torch.int32, device=self.device),
            torch.empty(n, self.row_width, dtype=torch.float32, device=self.device),
        )

    def _drain_evictions(self, ev):
        """Deferred: stash this batch's eviction log with an async count
        readback; the NEXT lookup drains it (avoids a device sync per batch —
        an evicted key re-looked-up within that one-batch window re-inits,
        which bounded-window LRU makes a cold-key non-event)."""
        keys_t, count_t, rows_t = ev
        if keys_t.numel() == 0:
            return
        self._drain_pending()
        cnt_pin = torch.zeros(1, dtype=torch.int32, pin_memory=True)
        cnt_pin.copy_(count_t, non_blocking=True)
        done = torch.cuda.Event()
        done.record()
        self._pending_evict = (keys_t, rows_t, cnt_pin, done)

    def _drain_pending(self):
        pend = getattr(self, "_pending_evict", None)
        if pend is None:
            return
        self._pending_evict = None
        keys_t, rows_t, cnt_pin, done = pend
        done.synchronize()
        cnt = int(cnt_pin.item())
        if cnt == 0:
            return
        self.evicted_total += cnt
        keys_np = keys_t[:cnt].cpu().numpy().view(np.uint64)
        rows_np = rows_t[:cnt].cpu().numpy()
        self.spill.insert(keys_np, rows_np)

    def flush_spill(self):
        """Drain any stashed eviction log (checkpoint/export paths)."""
        if self.spill is not None:
            self._drain_pending()

    def _opt_params(self):
        o = self.optimizer
        if isinstance(o, SGD):
            return [o.lr, o.weight_decay, 0.0, 0.0, 0.0, 0.0]
        if isinstance(o, Adagrad):
            return [
                o.lr,
                o.g_square_momentum,
                o.eps,
                1.0 if o.vectorwise_shared else 0.0,
                0.0,
                0.0,
            ]
        if isinstance(o, Adam):
            return [o.lr, o.betas[0], o.betas[1], o.eps, 0.0, 0.0]
        raise ValueError(o.kind)

    def __len__(self) -> int:
        return int((self.keys != 0).sum().item())

    @property
    def probe_stream(self) -> "torch.cuda.Stream":
        """Side stream for the spill-restore front half: its backlog is only
        this batch's tiny probe work, so the miss-mask readback does not wait
        for the lookup stream's multi-batch queue (the dcn-spill producer
        stall — profiles/README round-1 finding)."""
        if self._probe_stream is None:
            self._probe_stream = torch.cuda.Stream(device=self.device)
        return self._probe_stream

    def spill_restore(self, keys: torch.Tensor,
                      u_count: Optional[torch.Tensor] = None) -> None:
        """Phase 1 of a training lookup with a spill tier, split out so the
        engine can run it on ``probe_stream``: probe for missing-but-spilled
        rows, fetch them from the host tier and import into HBM.  The
        following ``lookup`` call reuses this tick and skips its own phase 1.

        Probes may race in-flight inserts/evictions on the lookup stream;
        both races degrade to a cold-key re-init or a no-op fetch — the
        bounded-staleness tolerance the table already guarantees."""
        tick = self.next_tick()
        self._restore_tick = tick
        if not len(self.spill) or keys.numel() == 0:
            return
        uc = (
            u_count
            if u_count is not None
            else torch.empty(0, dtype=torch.int64, device=self.device)
        )
        self._drain_pending()
        slots = self._C.store_probe(self.keys, self.ticks, keys, tick, uc)
        miss_keys = keys[slots < 0]  # syncs ONLY the caller's (probe) stream
        if miss_keys.numel():
            miss_np = miss_keys.cpu().numpy().view(np.uint64)
            rows, found = self.spill.fetch(miss_np)
            if found.any():
                ev = self._evict_buffers(int(found.sum()))
                found_keys = torch.from_numpy(
                    miss_np[found].view(np.int64).copy()
                ).to(self.device)
                rows_t = torch.from_numpy(rows[found]).to(self.device)
                self._C.store_import(
                    self.keys, self.ticks, self.arena, found_keys, rows_t,
                    tick, *ev,
                )
                self._drain_evictions(ev)

    def lookup(
        self, keys: torch.Tensor, train: bool,
        u_count: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """``u_count``: optional device-side i64[1] valid-prefix length for
        nnz-padded ``keys`` (sync-free dedup) — the padding tail is the
        empty-key sentinel and is never probed or claimed."""
        restored = self._restore_tick is not None
        if restored:
            tick = self._restore_tick
            self._restore_tick = None
        else:
            tick = self.next_tick()
        n = keys.numel()
        uc = (
            u_count
            if u_count is not None
            else torch.empty(0, dtype=torch.int64, device=self.device)
        )
        ev = self._no_evict
        if self.spill is not None and train and n:
            self._drain_pending()
            ev = self._evict_buffers(n)
        if (not restored and self.spill is not None and train and n
                and len(self.spill)):
            # spill phase 1 inline (engine did not pre-restore)
            slots = self._C.store_probe(self.keys, self.ticks, keys, tick, uc)
            miss_keys = keys[slots < 0]
            if miss_keys.numel():
                miss_np = miss_keys.cpu().numpy().view(np.uint64)
                rows, found = self.spill.fetch(miss_np)
                if found.any():
                    # re-size the log: imports can evict too
                    ev = self._evict_buffers(n + int(found.sum()))
                    found_keys = torch.from_numpy(
                        miss_np[found].view(np.int64).copy()
                    ).to(self.device)
                    rows_t = torch.from_numpy(rows[found]).to(self.device)
                    self._C.store_import(
                        self.keys, self.ticks, self.arena, found_keys, rows_t,
                        tick, *ev,
                    )
        out = torch.empty(n, self.dim, dtype=torch.float32, device=self.device)
        lo, hi = self.hyper.emb_initialization
        self._C.store_lookup(
            self.keys,
            self.ticks,
            self.arena,
            keys,
            out,
            self.dim,
            int(train),
            tick,
            float(lo),
            float(hi),
            float(self.hyper.admit_probability),
            float(self.optimizer.state_init(self.dim)),
            self.opt_space,
            *ev,
            uc,
        )
        self._drain_evictions(ev)
        return out

    def lookup_wire(self, keys: torch.Tensor, train: bool,
                    wire_dtype: torch.dtype) -> torch.Tensor:
        """Padded-a2a owner lookup: key 0 entries (bucket padding) return
        zeros and never claim slots (kernel-level skip); output is the wire
        dtype directly (fused f16 cast in init_gather)."""
        tick = self.next_tick()
        n = keys.numel()
        out = torch.empty(n, self.dim, dtype=wire_dtype, device=self.device)
        lo, hi = self.hyper.emb_initialization
        self._C.store_lookup(
            self.keys, self.ticks, self.arena, keys, out, self.dim,
            int(train), tick, float(lo), float(hi),
            float(self.hyper.admit_probability),
            float(self.optimizer.state_init(self.dim)), self.opt_space,
            *self._no_evict,
            torch.empty(0, dtype=torch.int64, device=self.device),
        )
        return out

    def update_gradients(self, keys: torch.Tensor, grads: torch.Tensor) -> int:
        powers = self._adam_step_powers()
        b1p, b2p = powers if powers else (0.0, 0.0)
        self._C.store_update(
            self.keys,
            self.ticks,
            self.arena,
            keys,
            grads.float().contiguous(),
            self.dim,
            self._opt_code,
            self._opt_params(),
            float(b1p),
            float(b2p),
            float(self.hyper.weight_bound),
            self._skipped,
        )
        return 0  # skipped count is accumulated device-side (no hot-path sync)

    def skipped_count(self) -> int:
        return int(self._skipped[0].item())

    def nan_row_count(self) -> int:
        return int(self._skipped[1].item())

    def export_keys(self, signs: np.ndarray) -> Tuple[np.ndarray, np.ndarray]:
        ks = hashing.splitmix64(signs.astype(np.uint64))
        ks[ks == EMPTY_KEY] = _ZERO_REMAP
        keys_t = torch.from_numpy(ks.view(np.int64)).to(self.device)
        slots = self._C.store_probe(
            self.keys, self.ticks, keys_t, self.tick,
            torch.empty(0, dtype=torch.int64, device=self.device),
        )
        found = slots >= 0
        inner = self.arena[slots[found]].cpu().numpy()
        found_np = found.cpu().numpy()
        out_signs, out_inner = signs[found_np], inner
        if self.spill is not None and (~found_np).any():
            # a touched sign may have been evicted to the host tier between
            # update and flush; fetch + re-insert (peek) keeps it resident
            self.flush_spill()
            miss, miss_ks = signs[~found_np], ks[~found_np]
            rows, sp_found = self.spill.fetch(miss_ks)
            if sp_found.any():
                self.spill.insert(miss_ks[sp_found], rows[sp_found])
                out_signs = np.concatenate([out_signs, miss[sp_found]])
                out_inner = np.concatenate([out_inner, rows[sp_found]])
        return out_signs, out_inner

    def export_rows(self) -> Tuple[np.ndarray, np.ndarray]:
        self.flush_spill()
        occ = torch.nonzero(self.keys != 0, as_tuple=False).view(-1)
        keys = self.keys[occ].cpu().numpy().view(np.uint64)
        signs = hashing.splitmix64_inv(keys)
        inner = self.arena[occ].cpu().numpy()
        sp = self._spill_export()
        if sp is not None:
            signs = np.concatenate([signs, sp[0]])
            inner = np.concatenate([inner, sp[1]])
        return signs, inner

    def import_rows(self, signs: np.ndarray, inner: np.ndarray) -> None:
        inner = _adapt_row_width(
            inner, self.row_width, float(self.optimizer.state_init(self.dim))
        )
        ks = hashing.splitmix64(signs.astype(np.uint64))
        ks[ks == EMPTY_KEY] = _ZERO_REMAP
        keys_t = torch.from_numpy(ks.view(np.int64)).to(self.device)
        rows_t = torch.from_numpy(np.ascontiguousarray(inner, dtype=np.float32)).to(
            self.device
        )
        tick = self.next_tick()
        ev = self._evict_buffers(keys_t.numel())
        self._C.store_import(
            self.keys, self.ticks, self.arena, keys_t, rows_t, tick, *ev
        )
        self._drain_evictions(ev)

    def clear(self) -> None:
        self.keys.zero_()
        self.ticks.zero_()
        self.arena.zero_()


def make_store(
    dim: int,
    capacity: int,
    optimizer: Optimizer,
    hyper: EmbeddingConfig,
    device: torch.device,
    spill_capacity: int = 0,
) -> EmbeddingStoreBase:
    if device.type == "cuda":
        return HipEmbeddingStore(dim, capacity, optimizer, hyper, device, spill_capacity)
    return CpuEmbeddingStore(dim, capacity, optimizer, hyper, device, spill_capacity)
