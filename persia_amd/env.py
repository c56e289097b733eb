"""Environment/rank plumbing.

Mirrors ``persia/env.py`` (reference persia/env.py:16-132): two coordinate
systems — ``RANK``/``LOCAL_RANK``/``WORLD_SIZE`` for trainer (nn-worker)
processes, ``REPLICA_INDEX``/``REPLICA_SIZE`` for auxiliary roles (data
loaders).  In the MI355X build the trainer ranks are the only GPU processes
(one per GPU, launched by torchrun); loaders are optional side processes.
"""
import os
from typing import Optional


def _int_env(name: str) -> Optional[int]:
    val = os.environ.get(name)
    if val is None or val == "":
        return None
    return int(val)


def get_rank() -> int:
    """Global rank of this trainer process (0 if unset)."""
    rank = _int_env("RANK")
    return 0 if rank is None else rank


def get_local_rank() -> int:
    """Local (intra-node) rank of this trainer process (0 if unset)."""
    rank = _int_env("LOCAL_RANK")
    return 0 if rank is None else rank


def get_world_size() -> int:
    """Number of trainer processes (1 if unset)."""
    ws = _int_env("WORLD_SIZE")
    return 1 if ws is None else ws


def get_replica_index() -> int:
    """Replica index for non-trainer roles (data loaders)."""
    idx = _int_env("REPLICA_INDEX")
    return 0 if idx is None else idx


def get_replica_size() -> int:
    """Replica count for non-trainer roles (data loaders)."""
    size = _int_env("REPLICA_SIZE")
    return 1 if size is None else size


def get_master_addr() -> str:
    return os.environ.get("MASTER_ADDR", "127.0.0.1")


def get_master_port() -> int:
    return int(os.environ.get("MASTER_PORT", "29500"))


PERSIA_SKIP_CHECK_DATA = os.environ.get("PERSIA_SKIP_CHECK_DATA", "false").lower() in (
    "1",
    "true",
    "yes",
)
