"""Sign hashing / routing math shared by the CPU reference path and tests.

The GPU path implements the identical functions in HIP
(persia_amd/csrc/common.h).  Where the reference uses farmhash64
(embedding_worker_service/mod.rs:342-345) we use splitmix64: it is a
*bijective* 64-bit mixer, which buys two MI355X-native properties:

* unique-by-hash == unique-by-sign, so the dedup sort can operate directly
  on hashed keys and range-partition them contiguously by owner rank;
* hashed keys can be inverted back to signs (no need to carry both).
"""
import numpy as np

_MASK = (1 << 64) - 1


def splitmix64(x: np.ndarray) -> np.ndarray:
    """Vectorized splitmix64 finalizer (bijective u64 -> u64)."""
    z = (x.astype(np.uint64) + np.uint64(0x9E3779B97F4A7C15)) & np.uint64(_MASK)
    z = (z ^ (z >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9) & np.uint64(_MASK)
    z = (z ^ (z >> np.uint64(27))) * np.uint64(0x94D049BB133111EB) & np.uint64(_MASK)
    return (z ^ (z >> np.uint64(31))) & np.uint64(_MASK)


def splitmix64_inv(z: np.ndarray) -> np.ndarray:
    """Exact inverse of :func:`splitmix64`."""
    z = z.astype(np.uint64)
    z = z ^ (z >> np.uint64(31)) ^ (z >> np.uint64(62))
    z = (z * np.uint64(0x319642B2D24D8EC3)) & np.uint64(_MASK)  # inv(0x94D049BB133111EB)
    z = z ^ (z >> np.uint64(27)) ^ (z >> np.uint64(54))
    z = (z * np.uint64(0x96DE1B173F119089)) & np.uint64(_MASK)  # inv(0xBF58476D1CE4E5B9)
    z = z ^ (z >> np.uint64(30)) ^ (z >> np.uint64(60))
    return (z - np.uint64(0x9E3779B97F4A7C15)) & np.uint64(_MASK)


def owner_of(hashed: np.ndarray, world_size: int) -> np.ndarray:
    """Range-partition hashed keys to ranks: owner = floor(h / 2^64 * W).

    Monotone in ``h`` for any W, so an h-sorted array is contiguous per
    owner (the xGMI all-to-all send buffer needs no extra shuffle)."""
    if world_size == 1:
        return np.zeros(len(hashed), dtype=np.int64)
    # avoid u128: use float is unsafe; do ((h >> 32) * W) >> 32 — monotone
    # partitioning on the top 32 bits (W <= 2^32).
    hi = (hashed.astype(np.uint64) >> np.uint64(32)).astype(np.uint64)
    return ((hi * np.uint64(world_size)) >> np.uint64(32)).astype(np.int64)


def apply_prefix(signs: np.ndarray, index_prefix: int, feature_spacing: int) -> np.ndarray:
    """Fold a slot's raw ids into its feature-group's prefixed sign space
    (reference: indices_add_prefix, embedding_worker_service/mod.rs:403-429)."""
    if index_prefix == 0:
        return signs.astype(np.uint64)
    return (signs.astype(np.uint64) % np.uint64(feature_spacing)) + np.uint64(index_prefix)


def hash_stack(signs: np.ndarray, rounds: int, embedding_size: int) -> np.ndarray:
    """Multi-round hash folding: each raw id becomes ``rounds`` bucketed ids
    (reference: indices_to_hashstack_indices, mod.rs:348-400, with farmhash
    replaced by splitmix64).  Returns shape (rounds, n)."""
    out = np.empty((rounds, len(signs)), dtype=np.uint64)
    h = signs.astype(np.uint64)
    for r in range(rounds):
        h = splitmix64(h)
        out[r] = h % np.uint64(embedding_size) + np.uint64(r * embedding_size)
    return out


def init_seed_for(sign: np.ndarray) -> np.ndarray:
    """Seed for deterministic per-sign row init (reference seeds SmallRng by
    sign, emb_entry.rs:35 — makes init independent of lookup order/sharding)."""
    return splitmix64(sign ^ np.uint64(0xA076_1D64_78BD_642F))
