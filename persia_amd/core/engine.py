"""The embedding engine: dedup → route → lookup → postprocess → backward.

This is the MI355X-native collapse of three reference tiers into one
per-GPU object:

* nn-worker forward/backward engines (rust/persia-core/src/forward.rs,
  backward.rs) — prefetch pipeline, staleness, H2D/D2H;
* embedding worker (embedding_worker_service/mod.rs) — dedup, sharding,
  summation/raw postprocess, gradient merge;
* embedding parameter server (embedding_parameter_service/mod.rs) — the
  sharded table + sparse optimizers.

Per batch (one dim-group):
  1. sign prep: hashstack folding + feature-group prefix + splitmix64 mix
     (the mixed key is both the dedup sort key and the shard router);
  2. dedup: radix sort + unique over the *whole batch* (all slots at once —
     the reference dedups per-slot per-worker with CPU hashmaps);
  3. route: contiguous range-partition of the sorted unique keys by owner
     rank, RCCL all_to_all_single over xGMI for keys, then rows (f16 wire,
     like the reference's f16 EmbeddingBatch wire format);
  4. owner-side: set-associative HBM hash-table probe/insert (HIP kernel);
  5. postprocess: fused gather + segment-sum into per-slot (B, dim) f16
     tensors, or raw distinct+index tensors (exact reference contract);
  6. backward: reverse — segment grad scatter, all_to_all, fused
     optimizer update on the owner shard.
"""
import queue
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from persia_amd.core import hashing
from persia_amd.core.comm import DistContext
from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig
from persia_amd.core.store import make_store
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.data import PersiaBatch
from persia_amd.embedding.optim import Optimizer
from persia_amd.logger import get_default_logger
from persia_amd.ops import reference as R

_logger = get_default_logger("persia_amd.engine")

_FLIP = -(2 ** 63)  # XOR with sign bit: signed sort order == unsigned order


def _roctx_range(name):
    """Method decorator: named roctx range (rocprofv3 --marker-trace) when
    the engine runs with PA_ROCTX=1; a bare call otherwise."""
    import functools

    def deco(fn):
        @functools.wraps(fn)
        def wrap(self, *a, **k):
            if not getattr(self, "_roctx", False):
                return fn(self, *a, **k)
            torch.cuda.nvtx.range_push(name)
            try:
                return fn(self, *a, **k)
            finally:
                torch.cuda.nvtx.range_pop()
        return wrap
    return deco


def _owner_of_keys(keys: torch.Tensor, world_size: int) -> torch.Tensor:
    """Monotone range-partition of u64 bit-pattern keys (int64 tensors) to
    ranks — same math as hashing.owner_of."""
    hi = (keys >> 32) & 0xFFFFFFFF
    return (hi * world_size) >> 32


def _dedup(keys_t: torch.Tensor):
    """Sort-based dedup of u64-bit-pattern keys.

    -> (uniq [U] ascending in u64 order, inverse [nnz] pos->uniq idx,
        perm [nnz] sort permutation, ustarts [U+1] uniq boundaries in sorted
        order).  perm/ustarts drive the ordered (deterministic, atomic-free)
        gradient scatter kernel: a unique key's positions are contiguous in
        sort order.  torch.sort on GPU is a hipCUB radix sort.
    """
    dev = keys_t.device
    nnz = keys_t.numel()
    if nnz == 0:
        z = torch.zeros(0, dtype=torch.int64, device=dev)
        return z, z, z, torch.zeros(1, dtype=torch.int64, device=dev)
    flipped = keys_t ^ _FLIP  # signed sort order == unsigned key order
    svals, perm = torch.sort(flipped)
    neq = torch.ones(nnz, dtype=torch.bool, device=dev)
    neq[1:] = svals[1:] != svals[:-1]
    inv_sorted = torch.cumsum(neq, 0) - 1
    inverse = torch.empty(nnz, dtype=torch.int64, device=dev)
    inverse[perm] = inv_sorted
    uniq = svals[neq] ^ _FLIP
    ustarts = torch.cat(
        [
            torch.nonzero(neq, as_tuple=False).view(-1),
            torch.tensor([nnz], dtype=torch.int64, device=dev),
        ]
    )
    return uniq, inverse, perm, ustarts


@dataclass
class SlotPayload:
    """What the dense side consumes for one slot (reference
    FeatureEmbeddingBatch, persia-common/src/lib.rs:85-113)."""

    name: str
    cfg: SlotConfig
    sum_tensor: Optional[torch.Tensor] = None  # [B, dim] f16
    raw_distinct: Optional[torch.Tensor] = None  # [U_slot+1, dim] f16, row0 pad
    raw_index: Optional[torch.Tensor] = None  # [B*sfs] i64
    raw_non_empty_index: Optional[torch.Tensor] = None
    raw_sample_id_num: Optional[torch.Tensor] = None

    @property
    def is_raw(self) -> bool:
        return self.raw_distinct is not None


@dataclass
class _SlotCtx:
    name: str
    cfg: SlotConfig
    pos_slice: Tuple[int, int]  # slice of group position space
    seg_offsets: torch.Tensor  # [B+1] per-slot CSR (into its own positions)
    sum_seg_base: int = -1  # sum slots: base index in the group segment space
    slot_uniq_global: Optional[torch.Tensor] = None  # raw slots: local->global uniq
    slot_inverse: Optional[torch.Tensor] = None  # raw slots: pos->local distinct


@dataclass
class _GroupCtx:
    dim: int
    uniq_keys: torch.Tensor  # [U] int64 (u64 bit pattern); fast path: [nnz]
    #   padded with the true count ONLY in u_count (device) — sync-free dedup
    inverse: torch.Tensor  # [nnz] i64
    perm: torch.Tensor  # [nnz] sort permutation (ordered grad scatter)
    ustarts: torch.Tensor  # [U+1] unique boundaries in sorted order
    u_count: Optional[torch.Tensor] = None  # [1] i64 device unique count
    slots: List[_SlotCtx] = field(default_factory=list)
    # fused sum-slot segment space (sum slots occupy positions [0, sum_end))
    cat_offsets: Optional[torch.Tensor] = None  # [n_sum_segs+1]
    seg_id: Optional[torch.Tensor] = None  # [nnz]: pos -> group seg idx, -1 raw
    seg_lens: Optional[torch.Tensor] = None  # [n_sum_segs] f32
    sqrt_mask: Optional[torch.Tensor] = None  # [n_sum_segs] bool
    sum_base: Optional[torch.Tensor] = None  # [n_sum_segs, dim] f16 fused output
    n_sum_slots: int = 0
    send_counts: Optional[List[int]] = None
    recv_counts: Optional[List[int]] = None
    # capacity-padded even-a2a fast path (sync-free distributed lookup):
    a2a_plan: Optional["_GroupPlan"] = None
    a2a_idx: Optional[torch.Tensor] = None  # [len(uniq)] uniq -> send-slot map
    a2a_recv_keys: Optional[torch.Tensor] = None  # [world*cap] owner-side keys
    a2a_owner_dedup: Optional[tuple] = None  # GPU: dedup_padded(recv_keys)


class _GroupPlan:
    """Static per-(group, batch-size) device constants for the all-single-ID
    fast path (the DLRM shape: every slot one id per sample).  Everything here
    is shape-derived, so one plan serves every batch of that shape."""

    def __init__(self, dim: int, names, prefixes_np: np.ndarray, B: int,
                 device: torch.device):
        S = len(names)
        self.dim = dim
        self.names = names
        self.B = B
        self.S = S
        self.slot_starts = torch.arange(0, (S + 1) * B, B, dtype=torch.int64,
                                        device=device)
        self.prefixes = torch.from_numpy(prefixes_np.view(np.int64)).to(device)
        self.cat_offsets = torch.arange(S * B + 1, dtype=torch.int64, device=device)
        self.seg_id = torch.arange(S * B, dtype=torch.int64, device=device)
        self.empty_scale = torch.empty(0, dtype=torch.float32, device=device)
        # padded-a2a constants (attached lazily by EmbeddingEngine._a2a_setup)
        self.a2a_world: Optional[int] = None
        self.a2a_cap: Optional[int] = None
        self.a2a_ar: Optional[torch.Tensor] = None  # arange(nnz)
        self.a2a_ones: Optional[torch.Tensor] = None  # ones(nnz) count weights
        self.owner_seg_id: Optional[torch.Tensor] = None  # arange(world*cap)


class _SplitSlots(torch.autograd.Function):
    """Split a group's fused sum tensor into per-slot views with ONE backward
    op: autograd's native per-view SliceBackward would zero-fill and add a
    full-base-sized buffer per slot (2×n_slots kernels per step)."""

    @staticmethod
    def forward(ctx, base: torch.Tensor, n_slots: int, B: int):
        ctx.n_slots, ctx.B, ctx.dim = n_slots, B, base.shape[1]
        ctx.dev, ctx.dtype = base.device, base.dtype
        return tuple(base[i * B : (i + 1) * B] for i in range(n_slots))

    @staticmethod
    def backward(ctx, *grads):
        parts = [
            g
            if g is not None
            else torch.zeros(ctx.B, ctx.dim, dtype=ctx.dtype, device=ctx.dev)
            for g in grads
        ]
        return torch.cat(parts, dim=0), None, None


class PersiaTrainingBatch:
    """Device-side batch ready for the dense model (reference
    PersiaTrainingBatch, persia-core/src/forward.rs:256-331)."""

    def __init__(self):
        self._payloads: List[SlotPayload] = []
        # fast-path groups whose SlotPayloads have not been built yet:
        # (group, slot_names, B).  The flagship loop reads group.sum_base
        # directly (enable_training_views / bench graph path), so building
        # ~2*n_slots python objects per batch in the pipeline thread is
        # wasted work unless someone actually asks for .payloads.
        self._lazy_sum_groups: List[Tuple["_GroupCtx", List[str], int]] = []
        self._order: Optional[Dict[str, int]] = None
        self._sorted: bool = False
        self.non_id_type_tensors: List[torch.Tensor] = []
        self.label_tensors: List[torch.Tensor] = []
        self.batch_size: int = 0
        self.requires_grad: bool = True
        self.meta: Optional[bytes] = None
        self.batch_id: Optional[int] = None
        self._groups: List[_GroupCtx] = []
        self._engine: Optional["EmbeddingEngine"] = None

    @property
    def payloads(self) -> List[SlotPayload]:
        """Per-slot payloads in the original id_type_features order
        (materialized on first access)."""
        if self._lazy_sum_groups:
            for group, _names, B in self._lazy_sum_groups:
                sums = group.sum_base
                for i, sc in enumerate(group.slots):
                    self._payloads.append(
                        SlotPayload(name=sc.name, cfg=sc.cfg,
                                    sum_tensor=sums[i * B : (i + 1) * B])
                    )
            self._lazy_sum_groups.clear()
            self._sorted = False
        if not self._sorted:
            if self._order is not None:
                self._payloads.sort(key=lambda p: self._order[p.name])
            self._sorted = True
        return self._payloads

    def enable_training_views(self) -> Dict[str, torch.Tensor]:
        """Mark each group's sum base requires_grad and return fresh per-slot
        views connected to it (autograd accumulates into ``sum_base.grad`` —
        consumed by ``EmbeddingEngine.apply_gradients_base``)."""
        views: Dict[str, torch.Tensor] = {}
        for group in self._groups:
            if group.sum_base is None:
                continue
            group.sum_base.requires_grad_(True)
            sum_names = [sc.name for sc in group.slots if sc.cfg.embedding_summation]
            parts = _SplitSlots.apply(group.sum_base, len(sum_names), self.batch_size)
            views.update(zip(sum_names, parts))
        return views

    def training_embeddings(self) -> List[torch.Tensor]:
        """payload-ordered embedding tensors for a manual train loop."""
        views = self.enable_training_views()
        return [views.get(p.name, p.sum_tensor) for p in self.payloads]

    def record_stream(self, stream: "torch.cuda.Stream") -> None:
        """Tag every device tensor produced on the lookup stream as in-use by
        ``stream`` so the caching allocator cannot recycle the memory for new
        lookup-stream allocations while consumer kernels still read it
        (mandatory for cross-stream tensor hand-off)."""
        for t in self._device_tensors():
            t.record_stream(stream)

    def _device_tensors(self):
        for t in self.non_id_type_tensors + self.label_tensors:
            if t.is_cuda:
                yield t
        for g in self._groups:
            a2a_owner = g.a2a_owner_dedup or ()
            for t in (g.uniq_keys, g.inverse, g.perm, g.ustarts, g.u_count,
                      g.sum_base, g.cat_offsets, g.seg_id, g.seg_lens,
                      g.sqrt_mask, g.a2a_idx, g.a2a_recv_keys, *a2a_owner):
                if t is not None and t.is_cuda:
                    yield t
            for sc in g.slots:
                for t in (sc.seg_offsets, sc.slot_uniq_global, sc.slot_inverse):
                    if t is not None and t.is_cuda:
                        yield t
        # lazy groups hold only sum payloads (no per-payload device tensors),
        # so iterating the eagerly-built list is sufficient and avoids
        # materializing payloads in the pipeline thread
        for p in self._payloads:
            for t in (p.raw_distinct, p.raw_index, p.raw_non_empty_index,
                      p.raw_sample_id_num):
                if t is not None and t.is_cuda:
                    yield t


class EmbeddingEngine:
    def __init__(
        self,
        schema: EmbeddingSchema,
        hyper: EmbeddingConfig,
        optimizer: Optional[Optimizer],
        gconf: GlobalConfig,
        device: torch.device,
        dist_ctx: Optional[DistContext] = None,
        wire_dtype: torch.dtype = torch.float16,
    ):
        self.schema = schema
        self.hyper = hyper
        self.optimizer = optimizer
        self.gconf = gconf
        self.device = device
        self.dist = dist_ctx or DistContext.from_default_group()
        # Lookups are issued from the pipeline THREAD and gradient pushes from
        # the main thread; each needs its OWN communicator so cross-thread
        # interleaving can't reorder collectives across ranks (NCCL requires
        # identical issue order per communicator on every rank).
        self.dist_grad = (
            DistContext.new_sparse_group() if self.dist.distributed else self.dist
        )
        self.wire_dtype = wire_dtype
        # group slots by dim (slots of one dim share one store + one a2a)
        self.groups: Dict[int, List[SlotConfig]] = {}
        for name in schema.slot_names():
            cfg = schema.get_slot(name)
            self.groups.setdefault(cfg.dim, []).append(cfg)
        if optimizer is None:
            from persia_amd.embedding.optim import SGD

            optimizer = SGD(lr=0.0)
            self.optimizer = optimizer
        self.stores = {
            dim: make_store(
                dim, gconf.capacity, optimizer, hyper, device, gconf.spill_capacity
            )
            for dim in self.groups
        }
        self.skipped_grad_signs = 0
        self.nan_grad_batches = 0
        self._plans = {}
        self._empty_scale = None
        self._empty_count = None
        self._pin_rings: Dict[int, list] = {}
        self._pin_idx: Dict[int, int] = {}
        self.model_manager_status = "Idle"
        self.model_manager_progress = 0.0
        import os as _os

        self._prod_timing = _os.environ.get("PA_PROD_TIMING", "0") == "1"
        # timers skip the first batches: one-time ATen kernel-module loads
        # (~hundreds of ms) otherwise dominate the averages
        self._pt = {"prep": 0.0, "native": 0.0, "batch": 0.0, "n": 0,
                    "skip": 20}
        # fused distributed path (capacity-padded even a2a, sync-free);
        # PA_FUSED_DIST=0 falls back to the exact two-phase counts exchange,
        # PA_FORCE_DIST=1 exercises the padded machinery at world_size=1
        # (copy in place of the collectives) for single-GPU tests/profiling
        self._fused_dist = _os.environ.get("PA_FUSED_DIST", "1") == "1"
        self._force_dist = _os.environ.get("PA_FORCE_DIST", "0") == "1"
        # PA_ROCTX=1: named roctx ranges around the pipeline stages for
        # rocprofv3 --marker-trace (torch.cuda.nvtx IS roctx on ROCm);
        # off by default so the hot loop never pays the marker calls
        self._roctx = _os.environ.get("PA_ROCTX", "0") == "1"
        self._a2a_overflow = torch.zeros(1, dtype=torch.int64, device=device)
        from persia_amd.core.metrics import EngineMetrics

        self.metrics_enabled = bool(gconf.enable_metrics)
        self.metrics = EngineMetrics(self.metrics_enabled)
        # hot-path observability: sampled every PA_METRICS_EVERY batches so
        # the flagship loop never pays for timers/copies it does not use;
        # GPU-side values (event timings, device counters) are harvested one
        # sample late through pinned async copies — no stream syncs
        self._metrics_every = max(1, int(_os.environ.get("PA_METRICS_EVERY", "64")))
        self._batch_counter = 0
        self._update_counter = 0
        self._pending_metric = None
        self._skipped_last = (0, 0)
        self._evicted_last = 0
        self.monitor = None
        if self.metrics_enabled:
            from persia_amd.core.monitor import DistinctIdMonitor

            self.monitor = DistinctIdMonitor()
        self.incremental = None
        if gconf.enable_incremental_update:
            # config-driven (reference inc-update lib.rs:79-120: the train
            # side starts the manager from the global config)
            self.enable_incremental_update(
                gconf.incremental_dir, gconf.incremental_buffer_size
            )

    def enable_incremental_update(self, dst_dir: str,
                                  buffer_size: int = 1_000_000):
        """Attach an incremental-update dumper: every gradient update's
        touched signs are recorded automatically and packets flush when the
        dedup buffer fills (reference inc-update lib.rs:178-312)."""
        from persia_amd.core.incremental import IncrementalUpdateDumper

        self.incremental = IncrementalUpdateDumper(self, dst_dir, buffer_size)
        return self.incremental

    def _empty_f32(self) -> torch.Tensor:
        if self._empty_scale is None:
            self._empty_scale = torch.empty(0, dtype=torch.float32, device=self.device)
        return self._empty_scale

    def _empty_i64(self) -> torch.Tensor:
        if self._empty_count is None:
            self._empty_count = torch.empty(0, dtype=torch.int64, device=self.device)
        return self._empty_count

    @staticmethod
    def _materialize_group(group: "_GroupCtx") -> "_GroupCtx":
        """Resolve a padded sync-free dedup group (fast path: uniq/ustarts are
        nnz-padded, true count on-device) to exact host shapes.  Costs one
        stream sync — only rare paths (raw grads, f32/large-dim fallback,
        skipped-grad groups) need it; the fused kernels read the count from
        the device."""
        if group.u_count is None:
            return group
        U = int(group.u_count.item())
        group.uniq_keys = group.uniq_keys.narrow(0, 0, U)
        group.ustarts = group.ustarts.narrow(0, 0, U + 1)
        group.u_count = None
        return group

    # ------------------------------------------------------------- sign prep

    def _prepare_slot_keys(self, feat) -> Tuple[np.ndarray, np.ndarray]:
        """raw uint64 ids -> prepared (mixed keys u64[nnz'], offsets i64[B+1]).
        Applies hashstack folding then feature-group prefix then splitmix64
        (reference: mod.rs:348-429 + sign_to_shard hashing)."""
        cfg = self.schema.get_slot(feat.name)
        values, offsets = feat.values, feat.offsets
        rounds = cfg.hash_stack_rounds
        if rounds > 0:
            stacked = hashing.hash_stack(
                values, rounds, cfg.hash_stack_config.embedding_size
            )  # (rounds, nnz)
            # per-sample segments repeat each position `rounds` times;
            # interleave position-major so each sample's span stays contiguous:
            # new position layout = for each sample, [r0 ids..., r1 ids..., ...]
            B = len(offsets) - 1
            lens = offsets[1:] - offsets[:-1]
            new_vals = np.empty(len(values) * rounds, dtype=np.uint64)
            new_offsets = np.zeros(B + 1, dtype=np.int64)
            np.cumsum(lens * rounds, out=new_offsets[1:])
            for r in range(rounds):
                for b in range(B):
                    s, e = offsets[b], offsets[b + 1]
                    dst = new_offsets[b] + r * (e - s)
                    new_vals[dst : dst + (e - s)] = stacked[r, s:e]
            values, offsets = new_vals, new_offsets
        signs = hashing.apply_prefix(values, cfg.index_prefix, self.schema.feature_spacing)
        keys = hashing.splitmix64(signs)
        keys[keys == np.uint64(0)] = np.uint64(0xD1B54A32D192ED03)
        return keys, offsets

    # ---------------------------------------------------------------- lookup

    def _exchange_rows(self, group: _GroupCtx, train: bool) -> torch.Tensor:
        """uniq keys -> rows [U, dim] (wire dtype on GPU path, f32 on CPU)."""
        store = self.stores[group.dim]
        if not self.dist.distributed:
            rows = store.lookup(group.uniq_keys, train, u_count=group.u_count)
            return rows
        owner = _owner_of_keys(group.uniq_keys, self.dist.world_size)
        send_counts = torch.bincount(owner, minlength=self.dist.world_size).tolist()
        recv_counts = self.dist.all_to_all_lengths(send_counts, self.device)
        group.send_counts, group.recv_counts = send_counts, recv_counts
        recv_keys = self.dist.all_to_all(group.uniq_keys, send_counts, recv_counts)
        rows_local = store.lookup(recv_keys, train).to(self.wire_dtype)
        rows = self.dist.all_to_all(rows_local, recv_counts, send_counts)
        return rows

    # ------------------------------------- padded even-a2a (fused distributed)

    def _a2a_setup(self, plan: "_GroupPlan") -> Tuple[int, int]:
        """One-time per-plan constants for the capacity-padded a2a.

        Every rank sends each peer a fixed ``cap``-row bucket (padding key 0 =
        "no entry"), so the variable per-destination counts NEVER reach the
        host — no ``.tolist()`` syncs in the hot loop (the reference's RPC
        naturally carried lengths; RCCL ``all_to_all_single`` with even splits
        is the xGMI-native equivalent).  Keys are hash-mixed ⇒ per-owner
        counts concentrate at nnz/world (binomial, σ≈√(nnz/world)); the slack
        of max(512, per/8) puts overflow beyond ~20σ.  A one-time allreduce
        verifies all ranks agree on nnz (weak scaling keeps shapes equal)."""
        if plan.a2a_cap is not None:
            return plan.a2a_world, plan.a2a_cap
        world = self.dist.world_size
        nnz = plan.S * plan.B
        if world > 1:
            lo, hi = self.dist.allreduce_int_minmax(nnz)
            if lo != hi:
                raise RuntimeError(
                    f"padded-a2a fast path needs equal per-rank batch shapes "
                    f"(nnz min {lo} != max {hi}); set PA_FUSED_DIST=0"
                )
            per = max(1, nnz // world)
            cap = per + max(512, per // 8)
        else:
            cap = nnz  # forced single-rank exercise: exact fit
        plan.a2a_world = world
        plan.a2a_cap = cap
        plan.a2a_ar = torch.arange(nnz, dtype=torch.int64, device=self.device)
        plan.a2a_ones = torch.ones(nnz, dtype=torch.int64, device=self.device)
        plan.owner_seg_id = torch.arange(
            world * cap, dtype=torch.int64, device=self.device
        )
        return world, cap

    def _a2a_route(self, plan: "_GroupPlan", uniq: torch.Tensor,
                   u_count: Optional[torch.Tensor]):
        """uniq (sorted, possibly nnz-padded with key 0) -> (send_keys
        [world*cap + 1] with per-owner buckets zero-padded, idx [len(uniq)]
        mapping each unique to its send slot; invalid/overflow -> the dummy
        tail slot).  Pure device ops — sync-free."""
        world, cap = plan.a2a_world, plan.a2a_cap
        if self.device.type == "cuda":
            # fused route: one bounds kernel (the owner partition of the
            # sorted keys is a range partition -> binary searches) + one
            # packed scatter — replaces the ~10-dispatch torch chain below
            # (measured 0.51 ms/batch of producer issue time)
            from persia_amd.ops import native as _native

            send, idx = _native().a2a_route(
                uniq,
                u_count if u_count is not None else self._empty_i64(),
                world, cap, self._a2a_overflow,
            )
            return send, idx
        n = uniq.numel()
        ar = plan.a2a_ar[:n]
        owner = _owner_of_keys(uniq, world)
        if u_count is not None:
            # padding tail (key 0) routes to the invalid bucket `world`
            owner = torch.where(ar < u_count, owner,
                                torch.full_like(owner, world))
        # NOT torch.bincount: its CUDA kernel sizes the histogram via
        # input.max().item() — a hidden device sync that would stall the
        # producer thread behind the whole lookup-stream backlog
        counts = torch.zeros(world + 1, dtype=torch.int64, device=uniq.device)
        counts.scatter_add_(0, owner, plan.a2a_ones[:n])
        starts = torch.cumsum(counts, 0) - counts
        pos = ar - starts.gather(0, owner)
        dummy = world * cap
        bad = (owner >= world) | (pos >= cap)
        idx = torch.where(bad, torch.full_like(pos, dummy), owner * cap + pos)
        send = torch.zeros(dummy + 1, dtype=torch.int64, device=uniq.device)
        send.scatter_(0, idx, uniq)
        self._a2a_overflow.add_((counts[:world] > cap).sum())
        return send, idx

    @_roctx_range("persia:a2a_exchange")
    def _a2a_exchange_fwd(self, plan: "_GroupPlan", group: _GroupCtx,
                          train: bool):
        """Padded forward exchange: route keys -> even a2a -> owner lookup in
        wire dtype -> even a2a back.  Returns (rows_full [world*cap+1, dim]
        with a zero dummy tail row, idx).  Fills group.a2a_* for backward."""
        world, cap = self._a2a_setup(plan)
        if self._prod_timing:
            _tr = time.perf_counter()
        send_keys, idx = self._a2a_route(plan, group.uniq_keys, group.u_count)
        comm = self.dist
        if world > 1:
            recv_keys = comm.all_to_all_even(send_keys[: world * cap])
        else:
            recv_keys = send_keys[: world * cap].clone()
        if self._prod_timing and getattr(self, "_prod_timing_live", True):
            self._pt["route"] = self._pt.get("route", 0.0) + (
                time.perf_counter() - _tr
            )
            _tr = time.perf_counter()
        store = self.stores[group.dim]
        rows_local = store.lookup_wire(recv_keys, train, self.wire_dtype)
        if self._prod_timing and getattr(self, "_prod_timing_live", True):
            self._pt["lkw"] = self._pt.get("lkw", 0.0) + (
                time.perf_counter() - _tr
            )
        rows_full = torch.empty(
            world * cap + 1, group.dim, dtype=rows_local.dtype,
            device=self.device,
        )
        if world > 1:
            comm.all_to_all_even(rows_local, out=rows_full[: world * cap])
        else:
            rows_full[: world * cap].copy_(rows_local)
        rows_full[world * cap].zero_()
        group.a2a_plan = plan
        group.a2a_idx = idx
        group.a2a_recv_keys = recv_keys
        if train and self.device.type == "cuda":
            # owner-side dedup for the backward merge, issued HERE so the
            # sort overlaps the dense step on the pipeline stream (the fused
            # a2a_route freed the producer budget this costs);
            # _a2a_backward_native falls back lazily if absent (infer skips)
            from persia_amd.ops import native as _native

            group.a2a_owner_dedup = tuple(_native().dedup_padded(recv_keys))
        return rows_full, idx

    def check_a2a_overflow(self) -> int:
        """Batches whose per-owner bucket overflowed `cap` (their keys were
        dropped to the dummy slot: rows read as zeros, grads discarded).
        Non-zero means the slack heuristic failed — syncs the device."""
        return int(self._a2a_overflow.sum().item())

    # ---------------------------------------------------------- forward path

    @_roctx_range("persia:process_batch")
    def process_batch(self, batch: PersiaBatch, train: Optional[bool] = None) -> PersiaTrainingBatch:
        if self._prod_timing:
            if self._pt["skip"] > 0:
                self._pt["skip"] -= 1
                self._prod_timing_live = False
            else:
                self._prod_timing_live = True
            _tb0 = time.perf_counter()
        self._batch_counter += 1
        sample = (
            self.metrics_enabled
            and self._batch_counter % self._metrics_every == 1 % self._metrics_every
        )
        if sample:
            _ms_t0 = time.perf_counter()
            _ms_ev0 = None
            if self.device.type == "cuda":
                _ms_ev0 = torch.cuda.Event(enable_timing=True)
                _ms_ev0.record()
        train = batch.requires_grad if train is None else train
        if self.gconf.job_type == "infer":
            train = False
        out = PersiaTrainingBatch()
        out.batch_size = batch.batch_size
        out.requires_grad = batch.requires_grad
        out.meta = batch.meta
        out.batch_id = batch.batch_id
        out._engine = self

        dev = self.device
        for i, x in enumerate(batch.non_id_type_features):
            out.non_id_type_tensors.append(
                self._upload_aux(batch, ("nid", i), x.data)
            )
        for i, x in enumerate(batch.labels):
            out.label_tensors.append(self._upload_aux(batch, ("lab", i), x.data))

        feats_by_dim: Dict[int, List] = {}
        for feat in batch.id_type_features:
            cfg = self.schema.get_slot(feat.name)
            feats_by_dim.setdefault(cfg.dim, []).append(feat)

        for dim, feats in feats_by_dim.items():
            group = self._process_group(dim, feats, out, train, src_batch=batch)
            out._groups.append(group)
        # keep payloads in the original id_type_features order (applied
        # lazily when .payloads is first materialized)
        out._order = {f.name: i for i, f in enumerate(batch.id_type_features)}
        out._sorted = False
        if sample:
            self._record_lookup_metrics(batch, out, _ms_t0, _ms_ev0)
        if self._prod_timing and self._prod_timing_live:
            self._pt["batch"] += time.perf_counter() - _tb0
            self._pt["n"] += 1
        return out

    def _record_lookup_metrics(self, batch, out, t0: float, ev0) -> None:
        """Sampled hot-path observability (reference worker/PS gauges,
        embedding_worker_service/mod.rs:83-100, parameter mod.rs:27-79).
        Device-side values are harvested one sample late (events/pinned
        copies complete by then) so this never synchronizes a stream."""
        m = self.metrics
        # host producer time of this batch (prep + kernel issue + any waits)
        m.lookup_preprocess_time_cost_sec.set(time.perf_counter() - t0)
        # distinct-id estimates from the raw host-side id arrays
        # (reference monitor.rs:29-114 samples the same point: pre-dedup ids)
        if self.monitor is not None:
            for feat in batch.id_type_features:
                vals = getattr(feat, "values", None)
                if vals is not None and len(vals):
                    self.monitor.observe(feat.name, vals)
            for name, est in self.monitor.estimates().items():
                m.distinct_id_estimate.labels(name).set(est)
        # unique-indices rate: exact on CPU; via the deferred u_count pinned
        # copy on GPU (padded dedup keeps U device-side)
        nnz = sum(
            len(getattr(f, "values", ())) for f in batch.id_type_features
        )
        # harvest the PREVIOUS sample's deferred device-side values
        pend = self._pending_metric
        self._pending_metric = None
        if pend is not None:
            ev_a, ev_b, ucnt_pin, skip_pin, p_nnz, done = pend
            if done.query():
                if ev_a is not None and ev_b is not None:
                    try:
                        m.lookup_hashmap_time_cost_sec.set(
                            ev_a.elapsed_time(ev_b) / 1e3
                        )
                    except Exception:
                        pass
                if ucnt_pin is not None and p_nnz:
                    m.batch_unique_indices_rate.set(
                        float(ucnt_pin.item()) / p_nnz
                    )
                if skip_pin is not None:
                    miss, nan = int(skip_pin[0].item()), int(skip_pin[1].item())
                    lm, ln = self._skipped_last
                    if miss > lm:
                        m.gradient_id_miss_count.inc(miss - lm)
                    if nan > ln:
                        m.nan_grad_skipped.inc(nan - ln)
                    self._skipped_last = (miss, nan)
        if self.device.type == "cuda":
            ev1 = torch.cuda.Event(enable_timing=True)
            ev1.record()
            ucnt_pin = None
            for g in out._groups:
                if g.u_count is not None:
                    ucnt_pin = torch.empty(1, dtype=torch.int64, pin_memory=True)
                    ucnt_pin.copy_(g.u_count, non_blocking=True)
                    break
                # exact dedup: U known host-side without a sync
                m.batch_unique_indices_rate.set(
                    g.uniq_keys.numel() / max(1, nnz)
                )
            skip_pin = None
            store = next(iter(self.stores.values()))
            if hasattr(store, "_skipped"):
                skip_pin = torch.empty(2, dtype=torch.int32, pin_memory=True)
                skip_pin.copy_(store._skipped, non_blocking=True)
            done = torch.cuda.Event()
            done.record()
            self._pending_metric = (ev0, ev1, ucnt_pin, skip_pin, nnz, done)
        else:
            U = sum(g.uniq_keys.numel() for g in out._groups)
            m.batch_unique_indices_rate.set(U / max(1, nnz))
            m.nan_grad_skipped.inc(0)
        # eviction counter (host-side total maintained by the spill drain)
        ev_tot = sum(
            getattr(s, "evicted_total", 0) for s in self.stores.values()
        )
        if ev_tot > self._evicted_last:
            m.evicted_count.inc(ev_tot - self._evicted_last)
            self._evicted_last = ev_tot

    def prepare_host_batch(self, batch: PersiaBatch) -> None:
        """Build a batch's pinned staging buffers host-side, without touching
        the GPU streams.  The first visit of a batch otherwise pays
        hipHostMalloc + copy inside the pipeline thread — ms-scale when the
        host is still reclaiming a prior process's pinned pages, which
        dominates short timed runs.  Idempotent; safe to call from the data
        producer (the reference allocates its pinned pools at startup for the
        same reason, cuda/pinned_memory_pool.rs)."""
        if self.device.type != "cuda":
            return
        pc = getattr(batch, "_pinned_cache", None)
        if pc is None:
            pc = batch._pinned_cache = {}
        for kind, items in (("nid", batch.non_id_type_features),
                            ("lab", batch.labels)):
            for i, x in enumerate(items):
                if (kind, i) in pc:
                    continue
                src = torch.from_numpy(np.ascontiguousarray(x.data))
                t = torch.empty_like(src, pin_memory=True)
                t.copy_(src)
                pc[(kind, i)] = t
        feats_by_dim: Dict[int, List] = {}
        for feat in batch.id_type_features:
            cfg = self.schema.get_slot(feat.name)
            feats_by_dim.setdefault(cfg.dim, []).append(feat)
        for dim, feats in feats_by_dim.items():
            if dim in pc or not all(
                getattr(f, "is_single", False) for f in feats
            ):
                continue
            values_np = np.concatenate([f.values for f in feats])
            pin = torch.empty(len(values_np), dtype=torch.int64, pin_memory=True)
            pin.numpy()[:] = values_np.view(np.int64)
            pc[dim] = pin

    def _upload_aux(self, batch, cache_key, arr: np.ndarray) -> torch.Tensor:
        """Dense-feature/label upload: pageable H2D stalls the pipeline
        thread ~0.1-0.2ms per batch, so stage through a per-batch pinned
        buffer (repeat visits are a pure async copy)."""
        if self.device.type != "cuda":
            return torch.from_numpy(np.ascontiguousarray(arr)).to(self.device)
        pc = getattr(batch, "_pinned_cache", None)
        if pc is None:
            pc = batch._pinned_cache = {}
        t = pc.get(cache_key)
        if t is None:
            src = torch.from_numpy(np.ascontiguousarray(arr))
            t = torch.empty_like(src, pin_memory=True)
            t.copy_(src)
            pc[cache_key] = t
        return t.to(self.device, non_blocking=True)

    def _upload_values(self, dim: int, values_np: np.ndarray) -> torch.Tensor:
        """Pinned-ring H2D: copy the host ids into a pinned slot and issue an
        async copy (pageable uploads stall the lookup thread ~0.2ms/batch)."""
        n = len(values_np)
        ring = self._pin_rings.setdefault(dim, [])
        if not ring or ring[0][0].numel() < n:
            ring.clear()
            for _ in range(12):  # > staleness window
                ring.append(
                    (torch.empty(n, dtype=torch.int64, pin_memory=True),
                     torch.cuda.Event())
                )
            self._pin_idx[dim] = 0
        idx = self._pin_idx[dim]
        self._pin_idx[dim] = (idx + 1) % len(ring)
        buf, ev = ring[idx]
        ev.synchronize()  # previous H2D from this slot must be done
        buf.numpy()[:n] = values_np.view(np.int64)
        vals_t = buf[:n].to(self.device, non_blocking=True)
        ev.record()
        return vals_t

    def _process_group_fast(self, dim: int, feats, out: PersiaTrainingBatch,
                            train: bool, src_batch=None) -> _GroupCtx:
        """All-single-ID sum-slot fast path (flagship DLRM shape): one H2D
        upload, ~15 kernel launches, static device-side plan."""
        from persia_amd.ops import native as _native

        if self._prod_timing:
            _t0 = time.perf_counter()
        C = _native()
        dev = self.device
        B = len(feats[0].values)
        names = tuple(f.name for f in feats)
        plan = self._plans.get((dim, names, B))
        if plan is None:
            prefixes_np = np.array(
                [self.schema.get_slot(n).index_prefix for n in names], dtype=np.uint64
            )
            plan = _GroupPlan(dim, names, prefixes_np, B, dev)
            self._plans[(dim, names, B)] = plan
        # per-batch pinned staging (allocated once per PersiaBatch, like a
        # loader-side pinned pool): repeat visits are a pure async H2D
        def _upload():
            pcache = (
                getattr(src_batch, "_pinned_cache", None)
                if src_batch is not None else None
            )
            if pcache is not None and dim in pcache:
                return pcache[dim].to(dev, non_blocking=True)
            values_np = np.concatenate([f.values for f in feats])
            if src_batch is not None:
                if pcache is None:
                    pcache = src_batch._pinned_cache = {}
                pin = torch.empty(len(values_np), dtype=torch.int64, pin_memory=True)
                pin.numpy()[:] = values_np.view(np.int64)
                pcache[dim] = pin
                return pin.to(dev, non_blocking=True)
            return self._upload_values(dim, values_np)

        spacing = self.schema.feature_spacing
        spacing_arg = spacing if spacing < (1 << 63) else -1
        slot_ctxs = [
            _SlotCtx(
                name=f.name, cfg=self.schema.get_slot(f.name),
                pos_slice=(i * B, (i + 1) * B), seg_offsets=None,
                sum_seg_base=i * B,
            )
            for i, f in enumerate(feats)
        ]
        store = self.stores[dim]
        dist_fast = (
            (self.dist.distributed or self._force_dist)
            and self._fused_dist
            and store.spill is None
        )
        # spill-tier front half on a dedicated probe stream: the miss-mask
        # readback in spill_restore then waits only for this batch's tiny
        # probe work, not the lookup stream's multi-batch backlog (the
        # dcn-spill ~1.9 ms/batch producer stall — round-1 profiles finding)
        spill_front = (
            store.spill is not None and not self.dist.distributed and train
        )
        if not self.dist.distributed and store.spill is None and not self._force_dist:
            # whole lookup in ONE native call (C++ drives sign prep, dedup,
            # probe/insert, gather and the fused segment-sum)
            if self._prod_timing and self._prod_timing_live:
                self._pt["prep"] += time.perf_counter() - _t0
            if self._prod_timing:
                _t1 = time.perf_counter()
            lo, hi = self.hyper.emb_initialization
            vals_t = _upload()
            sums, uniq_keys, inverse, perm, ustarts, u_count = C.lookup_local(
                vals_t, plan.slot_starts, plan.prefixes, spacing_arg,
                plan.cat_offsets, plan.empty_scale,
                store.keys, store.ticks, store.arena, dim,
                1 if train else 0, store.next_tick(), float(lo), float(hi),
                float(self.hyper.admit_probability),
                float(self.optimizer.state_init(dim)), store.opt_space,
            )
            group = _GroupCtx(
                dim=dim, uniq_keys=uniq_keys, inverse=inverse, perm=perm,
                ustarts=ustarts, u_count=u_count, slots=slot_ctxs,
                cat_offsets=plan.cat_offsets, seg_id=plan.seg_id,
                n_sum_slots=plan.S,
            )
            group.sum_base = sums
            # defer SlotPayload construction (consumers on the fused path read
            # group.sum_base directly; ~2*n_slots python objects per batch)
            out._lazy_sum_groups.append((group, [sc.name for sc in slot_ctxs], B))
            if self._prod_timing:
                self._pt["native"] += time.perf_counter() - _t1
            return group
        else:
            if spill_front:
                ps = store.probe_stream
                cur = torch.cuda.current_stream()
                with torch.cuda.stream(ps):
                    vals_t = _upload()
                    keys_t = C.sign_prep(
                        vals_t, plan.slot_starts, plan.prefixes, spacing_arg
                    )
                    uniq_keys, inverse, perm, ustarts, u_count = C.dedup_padded(
                        keys_t
                    )
                    store.spill_restore(uniq_keys, u_count)
                cur.wait_stream(ps)
                for t in (vals_t, uniq_keys, inverse, perm, ustarts, u_count):
                    t.record_stream(cur)
            else:
                vals_t = _upload()
                keys_t = C.sign_prep(
                    vals_t, plan.slot_starts, plan.prefixes, spacing_arg
                )
                if dist_fast or not self.dist.distributed:
                    # sync-free padded dedup: both the fused distributed
                    # exchange and the single-GPU spill path read the unique
                    # count on-device
                    uniq_keys, inverse, perm, ustarts, u_count = C.dedup_padded(
                        keys_t
                    )
                else:
                    uniq_keys, inverse, perm, ustarts = _dedup(keys_t)
                    u_count = None
            group = _GroupCtx(
                dim=dim, uniq_keys=uniq_keys, inverse=inverse, perm=perm,
                ustarts=ustarts, u_count=u_count, slots=slot_ctxs,
                cat_offsets=plan.cat_offsets, seg_id=plan.seg_id,
                n_sum_slots=plan.S,
            )
            if dist_fast:
                # capacity-padded even a2a (no host count syncs anywhere);
                # compose the unpack gather into `inverse` so segment_sum
                # reads the recv buffer directly — no [nnz, dim] row gather
                if self._prod_timing and getattr(self, "_prod_timing_live", True):
                    self._pt["dedup"] = self._pt.get("dedup", 0.0) + (
                        time.perf_counter() - _t0
                    )
                    _tx = time.perf_counter()
                rows_full, idx = self._a2a_exchange_fwd(plan, group, train)
                if self._prod_timing and getattr(self, "_prod_timing_live", True):
                    self._pt["exch"] = self._pt.get("exch", 0.0) + (
                        time.perf_counter() - _tx
                    )
                    _tx = time.perf_counter()
                inverse2 = idx.gather(0, inverse)
                sums = C.segment_sum(
                    rows_full, inverse2, plan.cat_offsets, plan.empty_scale
                )
                if self._prod_timing and getattr(self, "_prod_timing_live", True):
                    self._pt["sum"] = self._pt.get("sum", 0.0) + (
                        time.perf_counter() - _tx
                    )
                group.sum_base = sums
                out._lazy_sum_groups.append(
                    (group, [sc.name for sc in slot_ctxs], B)
                )
                if self._prod_timing and self._prod_timing_live:
                    self._pt["native"] += time.perf_counter() - _t0
                return group
            rows = self._exchange_rows(group, train)
            sums = C.segment_sum(
                rows.contiguous(), inverse, plan.cat_offsets, plan.empty_scale
            )
            group.sum_base = sums
        for i, sc in enumerate(slot_ctxs):
            out._payloads.append(
                SlotPayload(name=sc.name, cfg=sc.cfg, sum_tensor=sums[i * B : (i + 1) * B])
            )
        return group

    def _process_group(self, dim: int, feats, out: PersiaTrainingBatch, train: bool,
                       src_batch=None) -> _GroupCtx:
        dev = self.device
        native = dev.type == "cuda"
        if native:
            from persia_amd.ops import native as _native

            C = _native()
            if all(getattr(f, "is_single", False) for f in feats) and all(
                self.schema.get_slot(f.name).embedding_summation
                and self.schema.get_slot(f.name).hash_stack_rounds == 0
                for f in feats
            ):
                return self._process_group_fast(dim, feats, out, train, src_batch)
        # sum slots first so they occupy a contiguous prefix of the position
        # space (one fused segment-sum launch for the whole group)
        feats = sorted(
            feats, key=lambda f: 0 if self.schema.get_slot(f.name).embedding_summation else 1
        )
        gpu_prep = native
        key_arrays = []
        offset_arrays = []
        slot_ctxs: List[_SlotCtx] = []
        in_lens: List[int] = []      # uploaded (pre-expansion) lengths
        rounds_list: List[int] = []  # hashstack rounds per slot (GPU prep)
        sizes_list: List[int] = []
        pos = 0  # OUTPUT (post-hashstack-expansion) position space
        for feat in feats:
            cfg = self.schema.get_slot(feat.name)
            r = cfg.hash_stack_rounds
            if gpu_prep:
                # raw ids; hashstack expansion + prefix + mix all on-GPU.
                # expanded seg offsets are exactly offsets*r (every segment
                # grows by the same factor)
                keys = feat.values
                offsets = feat.offsets * r if r > 0 else feat.offsets
                out_len = len(keys) * max(r, 1)
            else:
                keys, offsets = self._prepare_slot_keys(feat)
                out_len = len(keys)
            key_arrays.append(keys)
            offset_arrays.append(offsets)
            in_lens.append(len(keys))
            rounds_list.append(r if gpu_prep else 0)
            sizes_list.append(
                cfg.hash_stack_config.embedding_size if r > 0 else 1
            )
            sc = _SlotCtx(name=feat.name, cfg=cfg, pos_slice=(pos, pos + out_len),
                          seg_offsets=None)
            slot_ctxs.append(sc)
            pos += out_len

        all_keys = np.concatenate(key_arrays) if key_arrays else np.empty(0, np.uint64)
        # single H2D upload of the whole group's values + all offsets
        keys_t = torch.from_numpy(all_keys.view(np.int64)).to(dev, non_blocking=False)
        all_offs = np.concatenate(
            [o for o in offset_arrays] or [np.zeros(1, np.int64)]
        )
        all_offs_t = torch.from_numpy(all_offs).to(dev)
        o0 = 0
        for sc, offs in zip(slot_ctxs, offset_arrays):
            sc.seg_offsets = all_offs_t[o0 : o0 + len(offs)]
            o0 += len(offs)
        if gpu_prep:
            out_starts = torch.tensor(
                [sc.pos_slice[0] for sc in slot_ctxs] + [pos],
                dtype=torch.int64, device=dev,
            )
            prefixes = torch.from_numpy(
                np.array([sc.cfg.index_prefix for sc in slot_ctxs], dtype=np.uint64)
                .view(np.int64)
            ).to(dev)
            spacing = self.schema.feature_spacing
            spacing_arg = spacing if spacing < (1 << 63) else -1
            from persia_amd.ops import native as _native2

            if any(rounds_list):
                in_starts = torch.from_numpy(
                    np.concatenate([[0], np.cumsum(in_lens)]).astype(np.int64)
                ).to(dev)
                off_starts = torch.from_numpy(
                    np.concatenate(
                        [[0], np.cumsum([len(o) for o in offset_arrays])]
                    ).astype(np.int64)
                ).to(dev)
                keys_t = _native2().sign_prep_stack(
                    keys_t, in_starts, out_starts, prefixes,
                    torch.tensor(rounds_list, dtype=torch.int32, device=dev),
                    torch.tensor(sizes_list, dtype=torch.int64, device=dev),
                    off_starts, all_offs_t, spacing_arg, pos,
                )
            else:
                keys_t = _native2().sign_prep(
                    keys_t, out_starts, prefixes, spacing_arg
                )

        uniq_keys, inverse, perm, ustarts = _dedup(keys_t)
        group = _GroupCtx(
            dim=dim, uniq_keys=uniq_keys, inverse=inverse, perm=perm,
            ustarts=ustarts, slots=slot_ctxs,
        )
        # CPU mirror of the GPU fused distributed path: same capacity-padded
        # even-a2a routing math, exact (unpadded) dedup — this is what the
        # gloo multi-process tests exercise so the 8-GPU driver run's code
        # path has CPU coverage
        use_padded = (
            not native
            and (self.dist.distributed or self._force_dist)
            and self._fused_dist
            and self.stores[dim].spill is None
            and all(getattr(f, "is_single", False) for f in feats)
            and all(
                sc.cfg.embedding_summation and sc.cfg.hash_stack_rounds == 0
                for sc in slot_ctxs
            )
        )
        if use_padded:
            B = out.batch_size
            names = tuple(sc.name for sc in slot_ctxs)
            plan = self._plans.get((dim, names, B))
            if plan is None:
                prefixes_np = np.array(
                    [self.schema.get_slot(n).index_prefix for n in names],
                    dtype=np.uint64,
                )
                plan = _GroupPlan(dim, names, prefixes_np, B, dev)
                self._plans[(dim, names, B)] = plan
            rows_full, idx = self._a2a_exchange_fwd(plan, group, train)
            rows = rows_full[idx]
        else:
            rows = self._exchange_rows(group, train)  # [U, dim]

        # ---- fused sum-slot postprocess
        sum_slots = [sc for sc in slot_ctxs if sc.cfg.embedding_summation]
        if sum_slots:
            base = 0
            offs_parts, sqrt_parts = [], []
            for sc in sum_slots:
                s, _e = sc.pos_slice
                sc.sum_seg_base = base
                nseg = sc.seg_offsets.numel() - 1
                offs_parts.append(sc.seg_offsets[:-1] + s)
                sqrt_parts.append(
                    torch.full((nseg,), bool(sc.cfg.sqrt_scaling), dtype=torch.bool, device=dev)
                )
                base += nseg
            sum_end = sum_slots[-1].pos_slice[1]
            cat_offsets = torch.cat(
                offs_parts + [torch.tensor([sum_end], dtype=torch.int64, device=dev)]
            )
            group.cat_offsets = cat_offsets
            group.sqrt_mask = torch.cat(sqrt_parts)
            lens = (cat_offsets[1:] - cat_offsets[:-1]).float()
            group.seg_lens = lens
            n_sum_segs = base
            seg_id = torch.full((inverse.numel(),), -1, dtype=torch.int64, device=dev)
            seg_id[:sum_end] = torch.repeat_interleave(
                torch.arange(n_sum_segs, dtype=torch.int64, device=dev),
                (cat_offsets[1:] - cat_offsets[:-1]),
            )
            group.seg_id = seg_id
            fwd_scale = torch.where(
                group.sqrt_mask, lens.clamp(min=1.0).rsqrt(), torch.ones_like(lens)
            )
            if native:
                sums = C.segment_sum(rows.contiguous(), inverse, cat_offsets, fwd_scale)
            else:
                sums = (
                    R.segment_sum_rows(rows, inverse[:sum_end], cat_offsets, False, torch.float32)
                    * fwd_scale.unsqueeze(1)
                ).to(torch.float16)
            group.sum_base = sums
            group.n_sum_slots = len(sum_slots)
            b0 = 0
            for sc in sum_slots:
                nseg = sc.seg_offsets.numel() - 1
                p = SlotPayload(name=sc.name, cfg=sc.cfg, sum_tensor=sums[b0 : b0 + nseg])
                out._payloads.append(p)
                b0 += nseg

        # ---- raw slots (torch path both backends: not in the flagship loop)
        for sc in slot_ctxs:
            if sc.cfg.embedding_summation:
                continue
            s, e = sc.pos_slice
            inv_slot = inverse[s:e]
            slot_uniq, slot_inv = torch.unique(inv_slot, sorted=True, return_inverse=True)
            sc.slot_uniq_global = slot_uniq
            sc.slot_inverse = slot_inv
            scale = 1.0
            rounds = sc.cfg.hash_stack_rounds
            if sc.cfg.sqrt_scaling and rounds > 1:
                scale = 1.0 / float(np.sqrt(rounds))
            distinct, index, non_empty, num = R.raw_embedding_tensors(
                rows[slot_uniq], slot_inv, sc.seg_offsets, sc.cfg.sample_fixed_size,
                scale, torch.float16,
            )
            out._payloads.append(
                SlotPayload(
                    name=sc.name, cfg=sc.cfg, raw_distinct=distinct, raw_index=index,
                    raw_non_empty_index=non_empty, raw_sample_id_num=num,
                )
            )
        return group

    # --------------------------------------------------------- backward path

    @_roctx_range("persia:update")
    def apply_gradients(
        self,
        training_batch: PersiaTrainingBatch,
        grads: Dict[str, Optional[torch.Tensor]],
        loss_scale: float = 1.0,
    ) -> None:
        """grads: slot name -> grad tensor ((B,dim) f16 for sum slots,
        (U_slot, dim) f32 for raw slots — the contract of ctx._on_backward,
        reference persia/ctx.py:926-1005) or None (skipped)."""
        self._update_counter += 1
        usample = (
            self.metrics_enabled
            and self._update_counter % self._metrics_every
            == 1 % self._metrics_every
        )
        if usample:
            _ut0 = time.perf_counter()
        native = self.device.type == "cuda"
        if native:
            from persia_amd.ops import native as _native

            C = _native()
        for group in training_batch._groups:
            self._materialize_group(group)  # generic path needs exact U
            U = group.uniq_keys.numel()
            buf = torch.zeros(U, group.dim, dtype=torch.float32, device=self.device)
            any_grad = False
            sum_slots = [sc for sc in group.slots if sc.cfg.embedding_summation]
            # all-None slot grads: skip the fused launch entirely so a fully
            # skipped group never applies a zero-gradient optimizer step
            # (reference skip-the-slot semantics, mod.rs:731-746)
            if native and sum_slots and not any(
                grads.get(sc.name) is not None for sc in sum_slots
            ):
                sum_slots = []
            if native and sum_slots:
                # fused ordered scatter over ALL sum slots in one launch.
                # Per-slot NaN skip (reference mod.rs:731-746) without a host
                # sync: zero the slot's per-segment scale via a device flag
                # and sanitize the grads (NaN*0 would still be NaN).
                g_parts, flag_parts = [], []
                B = training_batch.batch_size
                for sc in sum_slots:
                    g = grads.get(sc.name)
                    if g is None:
                        g = torch.zeros(B, group.dim, dtype=torch.float16, device=self.device)
                        flag = torch.ones((), dtype=torch.bool, device=self.device)
                    else:
                        any_grad = True
                        flag = torch.isnan(g).any()
                    g_parts.append(g.to(torch.float16))
                    flag_parts.append(flag.expand(sc.seg_offsets.numel() - 1))
                grads_cat = torch.nan_to_num(torch.cat(g_parts, dim=0))
                skip = torch.cat(flag_parts)
                lens = group.seg_lens
                scale = torch.where(
                    group.sqrt_mask, lens.clamp(min=1.0).rsqrt(), torch.ones_like(lens)
                )
                if loss_scale != 1.0:
                    scale = scale / loss_scale
                scale = torch.where(skip, torch.zeros_like(scale), scale)
                C.grad_scatter(
                    grads_cat.contiguous(), group.perm, group.ustarts,
                    group.seg_id, scale.contiguous(), buf, 1,
                )
                any_grad = True  # kernels launched regardless (no host sync)
            for sc in group.slots:
                g = grads.get(sc.name)
                if g is None:
                    continue
                if native and sc.cfg.embedding_summation:
                    continue  # handled by the fused scatter above
                # NaN filter (reference mod.rs:731-746: skip the whole slot)
                if bool(torch.isnan(g).any()):
                    self.nan_grad_batches += 1
                    _logger.warning(f"nan found in gradient update of {sc.name}, skipping")
                    continue
                any_grad = True
                s, e = sc.pos_slice
                inv_slot = group.inverse[s:e]
                if sc.cfg.embedding_summation:
                    contrib = R.segment_grad_scatter(
                        g,
                        inv_slot,
                        sc.seg_offsets,
                        U,
                        scale_factor=loss_scale,
                        sqrt_scaling=sc.cfg.sqrt_scaling,
                    )
                    buf += contrib
                else:
                    gf = g.float()
                    if loss_scale != 1.0:
                        gf = gf / loss_scale
                    rounds = sc.cfg.hash_stack_rounds
                    if sc.cfg.sqrt_scaling and rounds > 0:
                        gf = gf / float(np.sqrt(rounds))
                    buf.index_add_(0, sc.slot_uniq_global, gf)
            if not any_grad:
                continue
            self._route_and_update(group, buf)
        if usample:
            self.metrics.update_gradient_time_cost_sec.set(
                time.perf_counter() - _ut0
            )

    def _a2a_backward_native(self, group: _GroupCtx, gbase: torch.Tensor,
                             seg_scale: torch.Tensor, store, C) -> None:
        """GPU fused distributed backward: per-uniq ordered grad reduction
        written directly into the padded a2a send layout (f16 wire), even
        all-to-all on the grad communicator, then owner-side fused
        dedup+scatter+optimizer.  No [U, dim] buffer, no host syncs."""
        plan = group.a2a_plan
        world, cap = plan.a2a_world, plan.a2a_cap
        send_g = torch.empty(
            world * cap + 1, group.dim, dtype=torch.float16, device=self.device
        )
        uc = group.u_count if group.u_count is not None else self._empty_i64()
        C.grad_scatter_idx(
            gbase.contiguous(), group.perm, group.ustarts, group.seg_id,
            seg_scale, group.a2a_idx, send_g, uc,
        )
        if world > 1:
            recv_g = self.dist_grad.all_to_all_even(send_g[: world * cap])
        else:
            recv_g = send_g[: world * cap]
        # positions whose recv key is 0 (bucket padding) were never written:
        # scatter_update skips them on the key, before reading the grads
        if group.a2a_owner_dedup is None:
            group.a2a_owner_dedup = tuple(C.dedup_padded(group.a2a_recv_keys))
        ou, _oinv, operm, oustarts, ou_count = group.a2a_owner_dedup
        powers = store._adam_step_powers()
        b1p, b2p = powers if powers else (0.0, 0.0)
        C.scatter_update(
            store.keys, store.ticks, store.arena, ou, recv_g, operm,
            oustarts, plan.owner_seg_id, self._empty_f32(), group.dim,
            store._opt_code, store._opt_params(), float(b1p), float(b2p),
            float(self.hyper.weight_bound), store._skipped, ou_count,
        )
        if self.incremental is not None:
            # owner-side recording: this rank's shard's touched signs
            self.incremental.record_keys(ou, ou_count)

    def _route_and_update_padded(self, group: _GroupCtx, buf: torch.Tensor) -> None:
        """CPU mirror of the padded backward (the gloo-tested twin of
        _a2a_backward_native): pack per-uniq grads into the fixed bucket
        layout, even a2a, owner-side merge, ONE optimizer application per
        sign (sum-first, deterministic — same semantics as
        _route_and_update)."""
        plan = group.a2a_plan
        world, cap = plan.a2a_world, plan.a2a_cap
        store = self.stores[group.dim]
        send_g = torch.zeros(
            world * cap + 1, group.dim, dtype=self.wire_dtype,
            device=self.device,
        )
        # buf is exact [U, dim] (a materialized group); idx may be nnz-padded
        # — its first U entries are the valid uniques
        send_g.index_copy_(
            0, group.a2a_idx[: buf.shape[0]], buf.to(self.wire_dtype)
        )
        if world > 1:
            recv_g = self.dist_grad.all_to_all_even(send_g[: world * cap])
        else:
            recv_g = send_g[: world * cap]
        rk = group.a2a_recv_keys
        m = rk != 0
        if not bool(m.any()):
            return
        uniq_f, inv = torch.unique(rk[m] ^ _FLIP, sorted=True, return_inverse=True)
        merged = torch.zeros(
            uniq_f.numel(), group.dim, dtype=torch.float32, device=self.device
        )
        merged.index_add_(0, inv, recv_g[m].float())
        uk = uniq_f ^ _FLIP
        self.skipped_grad_signs += store.update_gradients(uk, merged)
        if self.incremental is not None:
            self.incremental.record_keys(uk)

    def _route_and_update(self, group: _GroupCtx, buf: torch.Tensor) -> None:
        store = self.stores[group.dim]
        if group.a2a_idx is not None:
            return self._route_and_update_padded(group, buf)
        if not self.dist.distributed:
            self.skipped_grad_signs += store.update_gradients(group.uniq_keys, buf)
            if self.incremental is not None:
                self.incremental.record_keys(group.uniq_keys)
            return
        comm = self.dist_grad  # main-thread communicator (see __init__)
        send_counts, recv_counts = group.send_counts, group.recv_counts
        if send_counts is None:
            owner = _owner_of_keys(group.uniq_keys, comm.world_size)
            send_counts = torch.bincount(owner, minlength=comm.world_size).tolist()
            recv_counts = comm.all_to_all_lengths(send_counts, self.device)
        keys_recv = comm.all_to_all(group.uniq_keys, send_counts, recv_counts)
        grads_recv = comm.all_to_all(
            buf.to(self.wire_dtype), send_counts, recv_counts
        )
        # merge duplicate keys across source ranks: pre-aggregate into one
        # optimizer application per sign (the reference applies each rank's
        # RPC sequentially under a per-sign lock — summing first is the
        # synchronous-equivalent, deterministic, and race-free on GPU)
        uniq_f, inv = torch.unique(keys_recv ^ _FLIP, sorted=True, return_inverse=True)
        merged = torch.zeros(
            uniq_f.numel(), group.dim, dtype=torch.float32, device=self.device
        )
        merged.index_add_(0, inv, grads_recv.float())
        uk = uniq_f ^ _FLIP
        self.skipped_grad_signs += store.update_gradients(uk, merged)
        if self.incremental is not None:
            self.incremental.record_keys(uk)

    @_roctx_range("persia:update")
    def apply_gradients_base(
        self,
        training_batch: PersiaTrainingBatch,
        raw_grads: Optional[Dict[str, Optional[torch.Tensor]]] = None,
        loss_scale: float = 1.0,
        sum_base_grads: Optional[List[torch.Tensor]] = None,
    ) -> None:
        """Backward fast path: sum-slot gradients are read directly from
        ``group.sum_base.grad`` (the per-group base tensor that
        ``prepare_features`` marked requires_grad), so there is no per-slot
        cat/NaN-reduce cascade — one ordered scatter per group.

        ``raw_grads``: slot name -> (U_slot, dim) f32 for raw slots (the
        index_add_ de-dup output of ctx._on_backward)."""
        native = self.device.type == "cuda"
        self._update_counter += 1
        usample = (
            self.metrics_enabled
            and self._update_counter % self._metrics_every
            == 1 % self._metrics_every
        )
        if usample:
            _ut0 = time.perf_counter()
        if not native:
            # CPU: slice the base into the generic per-slot dict
            grads: Dict[str, Optional[torch.Tensor]] = dict(raw_grads or {})
            for group in training_batch._groups:
                gbase = (
                    group.sum_base.grad if group.sum_base is not None else None
                )
                b0 = 0
                for sc in group.slots:
                    if not sc.cfg.embedding_summation:
                        continue
                    nseg = training_batch.batch_size
                    grads[sc.name] = (
                        gbase[b0 : b0 + nseg] if gbase is not None else None
                    )
                    b0 += nseg
            try:
                return self.apply_gradients(training_batch, grads, loss_scale)
            finally:
                if usample:
                    self.metrics.update_gradient_time_cost_sec.set(
                        time.perf_counter() - _ut0
                    )

        from persia_amd.ops import native as _native

        C = _native()
        for gi, group in enumerate(training_batch._groups):
            buf = None  # allocated lazily: the fused path never needs it
            if sum_base_grads is not None:
                gbase = sum_base_grads[gi]
            else:
                gbase = group.sum_base.grad if group.sum_base is not None else None
            if gbase is not None:
                # NaN gradients flow into buf and are skipped ROW-wise by the
                # update kernel (store.update_gradients) — no isnan pass here
                has_sqrt = group.sqrt_mask is not None and bool(group.sqrt_mask.any())
                if loss_scale == 1.0 and not has_sqrt:
                    seg_scale = self._empty_f32()
                else:
                    n_segs = gbase.shape[0]
                    seg_scale = torch.full(
                        (n_segs,), 1.0 / loss_scale, device=self.device
                    )
                    if has_sqrt:
                        seg_scale = seg_scale * torch.where(
                            group.sqrt_mask,
                            group.seg_lens.clamp(min=1.0).rsqrt(),
                            torch.ones_like(group.seg_lens),
                        )
                store = self.stores[group.dim]
                if (
                    group.a2a_idx is not None
                    and not raw_grads
                    and hasattr(store, "_opt_code")
                    and self.wire_dtype == torch.float16
                ):
                    # fused distributed backward: indexed scatter straight
                    # into the a2a send layout, even a2a, owner-side fused
                    # dedup+optimizer — zero host syncs
                    self._a2a_backward_native(group, gbase, seg_scale, store, C)
                    continue
                if (
                    not self.dist.distributed
                    and not raw_grads
                    and hasattr(store, "_opt_code")  # HipEmbeddingStore
                ):
                    # fused native backward: scatter + optimizer in one call
                    powers = store._adam_step_powers()
                    b1p, b2p = powers if powers else (0.0, 0.0)
                    C.update_local(
                        gbase.contiguous(), group.perm, group.ustarts,
                        group.seg_id, seg_scale, group.uniq_keys,
                        store.keys, store.ticks, store.arena, group.dim,
                        store._opt_code, store._opt_params(), float(b1p),
                        float(b2p), float(self.hyper.weight_bound),
                        store._skipped,
                        group.u_count
                        if group.u_count is not None
                        else self._empty_i64(),
                    )
                    if self.incremental is not None:
                        self.incremental.record_keys(
                            group.uniq_keys, group.u_count
                        )
                    continue
                self._materialize_group(group)
                buf = torch.empty(
                    group.uniq_keys.numel(), group.dim,
                    dtype=torch.float32, device=self.device,
                )
                C.grad_scatter(
                    gbase.contiguous(), group.perm, group.ustarts,
                    group.seg_id, seg_scale, buf, 0,
                )
            else:
                if not raw_grads:
                    # no sum-base grad and no raw grads: skip the group (a
                    # zero-grad optimizer application would still move Adam /
                    # Adagrad state — reference skips the slot instead)
                    continue
                self._materialize_group(group)
                buf = torch.zeros(
                    group.uniq_keys.numel(), group.dim,
                    dtype=torch.float32, device=self.device,
                )
            # raw slots add on top (buf fully written above)
            if raw_grads:
                for sc in group.slots:
                    if sc.cfg.embedding_summation:
                        continue
                    g = raw_grads.get(sc.name)
                    if g is None:
                        continue
                    if bool(torch.isnan(g).any()):
                        self.nan_grad_batches += 1
                        continue
                    gf = g.float()
                    if loss_scale != 1.0:
                        gf = gf / loss_scale
                    rounds = sc.cfg.hash_stack_rounds
                    if sc.cfg.sqrt_scaling and rounds > 0:
                        gf = gf / float(np.sqrt(rounds))
                    buf.index_add_(0, sc.slot_uniq_global, gf)
            self._route_and_update(group, buf)
        if usample:
            # host-side issue+wait time of the whole gradient push (the
            # reference's update_gradient_time_cost_sec, mod.rs:83-100)
            self.metrics.update_gradient_time_cost_sec.set(
                time.perf_counter() - _ut0
            )

    # ------------------------------------------------------------ checkpoint

    # checkpoint status machine (reference model-manager lib.rs:63-69:
    # {Dumping(progress) | Loading(progress) | Idle | Failed})
    def _set_status(self, status: str, progress: float = 0.0):
        self.model_manager_status = status
        self.model_manager_progress = progress

    def dump(self, dst_dir: str, blocking: bool = True) -> None:
        from persia_amd.core.checkpoint import dump_embedding

        def run():
            try:
                dump_embedding(self, dst_dir)
                self._set_status("Idle", 100.0)
            except Exception as e:
                _logger.error(f"embedding dump failed: {e}")
                self._set_status("Failed")
                if blocking:
                    raise

        # status set BEFORE the worker starts: a wait_for_emb_dumping issued
        # right after a non-blocking dump must never observe a stale 'Idle'
        self._set_status("Dumping", 0.0)
        if blocking:
            run()
        else:
            threading.Thread(target=run, daemon=True, name="persia-ckpt-dump").start()

    def load(self, src_dir: str, blocking: bool = True) -> None:
        from persia_amd.core.checkpoint import load_embedding

        def run():
            try:
                load_embedding(self, src_dir)
                self._set_status("Idle", 100.0)
            except Exception as e:
                _logger.error(f"embedding load failed: {e}")
                self._set_status("Failed")
                if blocking:
                    raise

        self._set_status("Loading", 0.0)  # before the thread (see dump)
        if blocking:
            run()
        else:
            threading.Thread(target=run, daemon=True, name="persia-ckpt-load").start()

    def wait_for_emb_dumping(self, timeout: float = 600.0) -> None:
        """Poll until the dump completes (reference rpc.rs:211-241)."""
        import time as _time

        t0 = _time.time()
        while getattr(self, "model_manager_status", "Idle") == "Dumping":
            if _time.time() - t0 > timeout:
                raise TimeoutError("embedding dump did not finish")
            _time.sleep(0.05)
        if getattr(self, "model_manager_status", "Idle") == "Failed":
            raise RuntimeError("embedding dump failed")

    def wait_for_emb_loading(self, timeout: float = 600.0) -> None:
        import time as _time

        t0 = _time.time()
        while getattr(self, "model_manager_status", "Idle") == "Loading":
            if _time.time() - t0 > timeout:
                raise TimeoutError("embedding load did not finish")
            _time.sleep(0.05)
        if getattr(self, "model_manager_status", "Idle") == "Failed":
            raise RuntimeError("embedding load failed")

    def num_resident_rows(self) -> int:
        return sum(len(s) for s in self.stores.values())


class ForwardPipeline:
    """Async prefetch with bounded staleness (reference ForwardImpl,
    forward.rs:470-528: input channel → lookup workers gated by a staleness
    Semaphore → output channel; the permit is released when the batch's
    gradients have been pushed — backward.rs:341-343)."""

    def __init__(self, engine: EmbeddingEngine, staleness: int = 8, out_buffer: int = 8,
                 reorder: bool = False):
        self.engine = engine
        self.staleness = max(1, staleness)
        # reproducible mode with several data loaders: process batches in
        # batch_id order via a min-heap (reference PerisaDataOrderManager,
        # forward.rs:396-468)
        self.reorder = reorder
        self._heap: list = []
        self._next_id = 0
        self._sem = threading.Semaphore(self.staleness)
        self._in: "queue.Queue" = queue.Queue(maxsize=max(2, out_buffer))
        self._out: "queue.Queue" = queue.Queue(maxsize=max(2, out_buffer))
        self._thread: Optional[threading.Thread] = None
        self._stop = threading.Event()
        self._exc: Optional[BaseException] = None
        self.inflight = 0  # lookups ahead of their gradient push (staleness)
        self.prod_s = 0.0  # cumulative producer-thread process_batch seconds
        self.prod_n = 0

    def start(self):
        if self._thread is None:
            self._thread = threading.Thread(target=self._run, daemon=True, name="persia-forward")
            self._thread.start()

    def _run(self):
        # On GPU, lookups run on a dedicated HIP stream so the sparse pipeline
        # overlaps the dense fwd/bwd on the default stream.  A fresh thread
        # defaults to device 0 — bind it to this rank's GPU first.
        stream = None
        if self.engine.device.type == "cuda":
            torch.cuda.set_device(self.engine.device)
            # PA_SPARSE_PRIORITY=-1 raises the lookup stream's scheduling
            # priority: on presets where the sparse stream is the critical
            # path (spill/dim-8 tables) the dense replay otherwise starves it
            import os as _os

            prio = int(_os.environ.get("PA_SPARSE_PRIORITY", "0"))
            stream = torch.cuda.Stream(device=self.engine.device, priority=prio)
        import heapq

        pending_eof = False
        while not self._stop.is_set():
            item = None
            if self.reorder and self._heap and self._heap[0][0] == self._next_id:
                item = heapq.heappop(self._heap)[1]
                self._next_id += 1
            else:
                try:
                    got = self._in.get(timeout=0.1)
                except queue.Empty:
                    if pending_eof and not self._heap:
                        self._out.put(None)
                        break
                    continue
                if got is None:
                    if self.reorder and self._heap:
                        pending_eof = True  # drain the heap first
                        continue
                    self._out.put(None)
                    break
                if self.reorder and got.batch_id is not None:
                    if got.batch_id != self._next_id:
                        heapq.heappush(self._heap, (got.batch_id, got))
                        continue
                    self._next_id += 1
                item = got
            self._sem.acquire()
            self.inflight += 1
            if self.engine.metrics_enabled:
                self.engine.metrics.staleness.set(self.inflight)
            try:
                t0 = time.perf_counter()
                if stream is not None:
                    with torch.cuda.stream(stream):
                        tb = self.engine.process_batch(item)
                    ev = torch.cuda.Event()
                    ev.record(stream)
                    tb._ready_event = ev
                else:
                    tb = self.engine.process_batch(item)
                # host-side producer cost (kernel issue + any CPU work);
                # excludes GPU completion — see `prod_n`/`prod_s` for the avg
                self.prod_s += time.perf_counter() - t0
                self.prod_n += 1
                self._out.put(tb)
            except BaseException as e:  # propagate to consumer
                self._exc = e
                self._out.put(None)
                break

    def put(self, batch: PersiaBatch):
        self.start()
        self._in.put(batch)
        if self.engine.metrics_enabled:
            self.engine.metrics.num_pending_batches.set(self._in.qsize())

    def finish(self):
        self.start()
        self._in.put(None)

    def get(self, timeout: Optional[float] = None) -> Optional[PersiaTrainingBatch]:
        tb = self._out.get(timeout=timeout)
        if tb is None and self._exc is not None:
            raise self._exc
        if tb is not None and getattr(tb, "_ready_event", None) is not None:
            cur = torch.cuda.current_stream()
            cur.wait_event(tb._ready_event)
            tb.record_stream(cur)
        return tb

    def release_permit(self):
        """Called after the batch's gradients were applied (or the batch was
        dropped) — the staleness window slides."""
        self.inflight = max(0, self.inflight - 1)
        self._sem.release()

    def stop(self):
        """Deterministic shutdown: wake and join the worker so no background
        thread is inside a HIP call at interpreter teardown."""
        self._stop.set()
        self._sem.release()  # unblock a worker waiting on the staleness gate
        try:
            self._in.put_nowait(None)
        except queue.Full:
            pass
        if self._thread is not None:
            self._thread.join(timeout=5.0)
            self._thread = None
