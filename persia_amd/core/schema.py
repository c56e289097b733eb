"""Embedding schema: per-slot configuration and feature-group prefix math.

Mirrors the reference's server-side ``EmbeddingConfig`` yaml
(rust/persia-embedding-config/src/lib.rs:535-650):

* ``SlotConfig {dim, sample_fixed_size, embedding_summation, sqrt_scaling,
  hash_stack_config, index_prefix}``
* ``feature_index_prefix_bit`` (default 8): the top bits of each u64 sign
  carry a feature-group prefix so that several slots can share one ID space
  without collisions; prefixes are auto-assigned
  ``(group_idx + 1) << (64 - prefix_bit)`` (lib.rs:600-650).
* ``feature_groups``: slots that share one prefix (and therefore one table).

The MI355X build groups slots by (dim) into *stores* — each store is one
GPU-sharded hash table; the prefix keeps slots of a group disjoint inside it.
"""
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import yaml


@dataclass
class HashStackConfig:
    """Multi-round hash folding of an unbounded ID space into
    ``hash_stack_rounds × embedding_size`` buckets
    (reference: embedding_worker_service/mod.rs:348-400)."""

    hash_stack_rounds: int = 0
    embedding_size: int = 0


@dataclass
class SlotConfig:
    name: str
    dim: int
    sample_fixed_size: int = 10
    embedding_summation: bool = True
    sqrt_scaling: bool = False
    hash_stack_config: Optional[HashStackConfig] = None
    index_prefix: int = 0  # assigned by EmbeddingSchema

    @property
    def hash_stack_rounds(self) -> int:
        return self.hash_stack_config.hash_stack_rounds if self.hash_stack_config else 0


@dataclass
class EmbeddingSchema:
    """The full per-job embedding layout.

    ``feature_groups`` maps group name -> list of slot names. Slots without an
    explicit group each get their own group (reference behavior when
    ``feature_groups`` is empty: every slot is its own group).
    """

    slots: Dict[str, SlotConfig]
    feature_index_prefix_bit: int = 8
    feature_groups: Dict[str, List[str]] = field(default_factory=dict)

    def __post_init__(self):
        # Assign every slot to a group.
        grouped = set()
        for names in self.feature_groups.values():
            grouped.update(names)
        for name in self.slots:
            if name not in grouped:
                self.feature_groups[name] = [name]
        # Deterministic group order = insertion order of feature_groups.
        # Prefix assignment mirrors config lib.rs:600-650.
        if self.feature_index_prefix_bit > 0:
            for group_idx, (_gname, names) in enumerate(self.feature_groups.items()):
                prefix = (group_idx + 1) << (64 - self.feature_index_prefix_bit)
                if group_idx + 1 >= (1 << self.feature_index_prefix_bit):
                    raise ValueError(
                        f"too many feature groups ({len(self.feature_groups)}) for "
                        f"feature_index_prefix_bit={self.feature_index_prefix_bit}"
                    )
                for n in names:
                    self.slots[n].index_prefix = prefix

    @property
    def feature_spacing(self) -> int:
        """Non-prefix sign space per feature group
        (reference: embedding_worker_service/mod.rs:404-408)."""
        if self.feature_index_prefix_bit > 0:
            return (1 << (64 - self.feature_index_prefix_bit)) - 1
        return (1 << 64) - 1

    def slot_names(self) -> List[str]:
        return list(self.slots.keys())

    def get_slot(self, name: str) -> SlotConfig:
        return self.slots[name]

    @staticmethod
    def from_dict(d: dict) -> "EmbeddingSchema":
        slots = {}
        for name, sc in d.get("slots_config", d.get("slots", {})).items():
            hs = sc.get("hash_stack_config")
            hsc = (
                HashStackConfig(
                    hash_stack_rounds=hs.get("hash_stack_rounds", 0),
                    embedding_size=hs.get("embedding_size", 0),
                )
                if hs
                else None
            )
            slots[name] = SlotConfig(
                name=name,
                dim=int(sc["dim"]),
                sample_fixed_size=int(sc.get("sample_fixed_size", 10)),
                embedding_summation=bool(sc.get("embedding_summation", True)),
                sqrt_scaling=bool(sc.get("sqrt_scaling", False)),
                hash_stack_config=hsc,
            )
        return EmbeddingSchema(
            slots=slots,
            feature_index_prefix_bit=int(d.get("feature_index_prefix_bit", 8)),
            feature_groups={k: list(v) for k, v in d.get("feature_groups", {}).items()},
        )

    @staticmethod
    def from_yaml(path: str) -> "EmbeddingSchema":
        with open(path, "r", encoding="utf-8") as f:
            return EmbeddingSchema.from_dict(yaml.safe_load(f))

    def to_dict(self) -> dict:
        out = {
            "feature_index_prefix_bit": self.feature_index_prefix_bit,
            "slots_config": {},
            "feature_groups": {k: list(v) for k, v in self.feature_groups.items()},
        }
        for name, s in self.slots.items():
            sc = {
                "dim": s.dim,
                "sample_fixed_size": s.sample_fixed_size,
                "embedding_summation": s.embedding_summation,
                "sqrt_scaling": s.sqrt_scaling,
            }
            if s.hash_stack_config:
                sc["hash_stack_config"] = {
                    "hash_stack_rounds": s.hash_stack_config.hash_stack_rounds,
                    "embedding_size": s.hash_stack_config.embedding_size,
                }
            out["slots_config"][name] = sc
        return out


@dataclass
class GlobalConfig:
    """Job-level knobs (reference: PersiaGlobalConfig, config lib.rs:417-470).

    ``capacity`` is the max resident rows **per rank shard** (LRU-style
    eviction beyond it, like the reference parameter server's EvictionMap
    with default 1e9 rows; here bounded by HBM)."""

    job_type: str = "train"  # train | eval | infer
    capacity: int = 1 << 24
    spill_capacity: int = 0  # host-DRAM rows per rank (0 = spill disabled)
    buckets_pow2: bool = True
    enable_metrics: bool = False
    checkpointing_workers: int = 4
    staleness: int = 8
    reproducible: bool = False
    # online-inference freshness stream (reference config lib.rs:417-447:
    # enable_incremental_update / incremental_buffer_size / incremental_dir)
    enable_incremental_update: bool = False
    incremental_buffer_size: int = 1_000_000
    incremental_dir: str = "/workspace/inc_dir" 

    @staticmethod
    def from_yaml(path: str) -> "GlobalConfig":
        with open(path, "r", encoding="utf-8") as f:
            d = yaml.safe_load(f) or {}
        common = d.get("common_config", d)
        server = d.get("embedding_parameter_server_config", {})
        return GlobalConfig(
            job_type=str(common.get("job_type", "train")).lower(),
            capacity=int(server.get("capacity", common.get("capacity", 1 << 24))),
            enable_metrics=bool(
                common.get("metrics_config", {}).get("enable_metrics", False)
            ),
            checkpointing_workers=int(
                common.get("checkpointing_config", {}).get("num_workers", 4)
            ),
            spill_capacity=int(server.get("spill_capacity", 0)),
            staleness=int(common.get("embedding_staleness",
                                     common.get("staleness", 8))),
            reproducible=bool(common.get("reproducible", False)),
            enable_incremental_update=bool(
                server.get("enable_incremental_update", False)
            ),
            incremental_buffer_size=int(
                server.get("incremental_buffer_size", 1_000_000)
            ),
            incremental_dir=str(
                server.get("incremental_dir", "/workspace/inc_dir")
            ),
        )
