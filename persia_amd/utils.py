"""Misc utilities (mirrors reference persia/utils.py:13-91)."""
import os
import random
import socket
from typing import Any

import numpy as np

import yaml


def setup_seed(seed: int) -> None:
    """Seed python/numpy/torch and enable deterministic algorithms.

    Reference: persia/utils.py:13-31 — the deterministic-mode anchor used by
    the adult-income reproducibility gate.
    """
    import torch

    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    torch.use_deterministic_algorithms(True, warn_only=True)


def load_yaml(path: str) -> Any:
    with open(path, "r", encoding="utf-8") as f:
        return yaml.safe_load(f)


def dump_yaml(obj: Any, path: str) -> None:
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    with open(path, "w", encoding="utf-8") as f:
        yaml.safe_dump(obj, f)


def find_free_port() -> int:
    """Find a free TCP port (reference: persia/utils.py:76-91)."""
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        return s.getsockname()[1]
