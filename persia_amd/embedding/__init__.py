"""Embedding hyper-parameters (mirrors reference persia/embedding/__init__.py:1-26)."""
from dataclasses import dataclass
from typing import Tuple


@dataclass
class EmbeddingConfig:
    """Server-side embedding hyperparameters.

    Arguments mirror the reference ``persia.embedding.EmbeddingConfig``:

    * ``emb_initialization``: (lower, upper) of the bounded-uniform row init
      (seeded per sign for reproducibility, reference emb_entry.rs:35).
    * ``admit_probability``: probability that a first-seen sign is admitted
      into the table (reference embedding_parameter_service/mod.rs:205-213).
    * ``weight_bound``: post-update clamp to ±bound (persia-simd weight_bound).
    """

    emb_initialization: Tuple[float, float] = (-0.01, 0.01)
    admit_probability: float = 1.0
    weight_bound: float = 10.0
