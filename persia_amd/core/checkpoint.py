"""Embedding checkpoint dump/load.

Directory layout mirrors the reference model manager
(persia-model-manager/src/lib.rs:124-240):

    dst/
      s0/ replica_0_shard_0.emb ...
      s1/ ...
      embedding_dump_done          <- yaml marker {num_shards,
                                      num_internal_shards, datetime}

The reference's speedy-serialized ArrayLinkedList framing is replaced by a
documented little-endian record format (the speedy submodule is not
verifiable — SURVEY §5 Checkpoint note):

    file   := magic "PAEMB1\\0\\0" | u64 num_records | u64 embedding_dim
              | u64 row_width
              | u64 signs[num_records]                  (columnar: all signs)
              | f32 inner[num_records * row_width]      (then all rows)
    inner row = emb(dim) ‖ opt_state — reference emb_entry.rs:17-23 layout.
    All fields little-endian; the two arrays are contiguous blocks, NOT
    interleaved per record.

Re-sharding on load when the dumped shard count differs from the current
world size mirrors mod.rs:1150-1259: every rank scans all files and keeps the
signs it owns.
"""
import datetime
import os
import struct
from typing import TYPE_CHECKING, Tuple

import numpy as np
import yaml

from persia_amd.core import hashing
from persia_amd.logger import get_default_logger

if TYPE_CHECKING:
    from persia_amd.core.engine import EmbeddingEngine

_logger = get_default_logger("persia_amd.checkpoint")

_MAGIC = b"PAEMB1\x00\x00"
DONE_MARKER = "embedding_dump_done"


def write_emb_file(path: str, signs: np.ndarray, inner: np.ndarray, dim: int) -> None:
    n, row_width = inner.shape
    assert len(signs) == n
    with open(path, "wb") as f:
        f.write(_MAGIC)
        f.write(struct.pack("<QQQ", n, dim, row_width))
        f.write(np.ascontiguousarray(signs, dtype=np.uint64).tobytes())
        f.write(np.ascontiguousarray(inner, dtype=np.float32).tobytes())


def read_emb_file(path: str) -> Tuple[np.ndarray, np.ndarray, int]:
    with open(path, "rb") as f:
        magic = f.read(8)
        assert magic == _MAGIC, f"bad .emb magic in {path}"
        n, dim, row_width = struct.unpack("<QQQ", f.read(24))
        signs = np.frombuffer(f.read(8 * n), dtype=np.uint64).copy()
        inner = (
            np.frombuffer(f.read(4 * n * row_width), dtype=np.float32)
            .reshape(n, row_width)
            .copy()
        )
    return signs, inner, int(dim)


def dump_embedding(engine: "EmbeddingEngine", dst_dir: str) -> None:
    rank = engine.dist.rank
    shard_dir = os.path.join(dst_dir, f"s{rank}")
    os.makedirs(shard_dir, exist_ok=True)
    dims = sorted(engine.stores.keys())
    for shard_idx, dim in enumerate(dims):
        signs, inner = engine.stores[dim].export_rows()
        path = os.path.join(shard_dir, f"replica_{rank}_shard_{shard_idx}.emb")
        write_emb_file(path, signs, inner, dim)
    # replica-level marker
    with open(os.path.join(shard_dir, DONE_MARKER), "w", encoding="utf-8") as f:
        yaml.safe_dump(
            {
                "num_shards": engine.dist.world_size,
                "num_internal_shards": len(dims),
                "datetime": datetime.datetime.now().isoformat(),
            },
            f,
        )
    engine.dist_grad.barrier()
    if rank == 0:
        with open(os.path.join(dst_dir, DONE_MARKER), "w", encoding="utf-8") as f:
            yaml.safe_dump(
                {
                    "num_shards": engine.dist.world_size,
                    "num_internal_shards": len(dims),
                    "datetime": datetime.datetime.now().isoformat(),
                },
                f,
            )
    engine.dist_grad.barrier()


def load_embedding(engine: "EmbeddingEngine", src_dir: str) -> None:
    marker = os.path.join(src_dir, DONE_MARKER)
    assert os.path.exists(marker), f"no {DONE_MARKER} marker in {src_dir}"
    with open(marker, "r", encoding="utf-8") as f:
        info = yaml.safe_load(f)
    num_shards = int(info["num_shards"])
    rank, world = engine.dist.rank, engine.dist.world_size
    dims = sorted(engine.stores.keys())

    def import_file(path: str, filter_owner: bool):
        signs, inner, dim = read_emb_file(path)
        if dim not in engine.stores:
            _logger.warning(f"checkpoint dim {dim} has no configured slots; skipping {path}")
            return
        if filter_owner and world > 1:
            owner = hashing.owner_of(hashing.splitmix64(signs), world)
            keep = owner == rank
            signs, inner = signs[keep], inner[keep]
        if len(signs):
            engine.stores[dim].import_rows(signs, inner)

    if num_shards == world:
        shard_dir = os.path.join(src_dir, f"s{rank}")
        for fn in sorted(os.listdir(shard_dir)):
            if fn.endswith(".emb"):
                import_file(os.path.join(shard_dir, fn), filter_owner=False)
    else:
        # re-shard: scan every replica dir, keep owned signs
        for r in range(num_shards):
            shard_dir = os.path.join(src_dir, f"s{r}")
            for fn in sorted(os.listdir(shard_dir)):
                if fn.endswith(".emb"):
                    import_file(os.path.join(shard_dir, fn), filter_owner=True)
    engine.dist_grad.barrier()
