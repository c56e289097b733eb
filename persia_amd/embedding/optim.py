"""Sparse (embedding) optimizer configuration.

Mirrors the reference ``persia/embedding/optim.py`` API.  In the MI355X build
these are plain config dataclasses consumed by the in-process embedding
engine (the reference publishes them over NATS to remote parameter servers —
persia/embedding/optim.py + persia-common/src/optim.rs).

Update math is identical to the reference AVX2 kernels
(rust/persia-simd/src/lib.rs):

* SGD:      ``w -= lr * (g + wd * w)``                          (lib.rs:124)
* Adagrad:  ``w -= lr * g * rsqrt(acc + eps); acc = acc*m + g²`` (lib.rs:21;
  note the accumulator read happens *before* its update)
* Adam:     bias-corrected with per-feature-group β-power accumulators
  (persia-common/src/optim.rs:147-216)
"""
from abc import ABC
from typing import Tuple


class Optimizer(ABC):
    """Base embedding-optimizer config."""

    kind = "none"

    def require_space(self, dim: int) -> int:
        """Extra f32s of per-row optimizer state appended after the embedding
        (reference: Optimizable::require_space)."""
        return 0

    def state_init(self, dim: int) -> float:
        """Initial value of the optimizer-state region."""
        return 0.0

    def apply(self):
        """Reference-API compat: registering happens implicitly when the
        engine is created; kept as a no-op hook."""


class SGD(Optimizer):
    kind = "sgd"

    def __init__(self, lr: float, momentum: float = 0.0, weight_decay: float = 0.0):
        self.lr = lr
        self.momentum = momentum  # accepted, unused (reference parity: NaiveSGD has no momentum)
        self.weight_decay = weight_decay


class Adam(Optimizer):
    """Sparse Adam with bias correction via pre-computed beta powers.

    β-power bookkeeping (documented divergence from the reference,
    persia-common/src/optim.rs:147-216): the reference keeps one
    (β1ᵗ, β2ᵗ) accumulator PER FEATURE GROUP (keyed by the masked sign
    prefix) and advances it when that group receives an update batch.  Here
    the accumulator lives per dim-group STORE and advances once per update
    call.  The two are identical whenever every feature group appears in
    every batch — the invariant of this architecture's dim-group batching
    and of every shipped example/bench config.  They diverge only for jobs
    that send partial batches (some slots absent), where a group skipping a
    batch here still sees its bias-correction step advance: a slightly
    smaller early-step correction, not a correctness issue (both schemes
    converge to the same asymptotic update).  CPU and HIP backends implement
    the same store-level scheme bit-identically (oracle-tested).
    """

    kind = "adam"

    def __init__(
        self,
        lr: float = 1e-3,
        betas: Tuple[float, float] = (0.9, 0.999),
        weight_decay: float = 0.0,
        eps: float = 1e-8,
    ):
        self.lr = lr
        self.betas = betas
        self.weight_decay = weight_decay  # accepted, unused by reference impl
        self.eps = eps

    def require_space(self, dim: int) -> int:
        return 2 * dim


class Adagrad(Optimizer):
    kind = "adagrad"

    def __init__(
        self,
        lr: float = 1e-2,
        initial_accumulator_value: float = 1e-2,
        weight_decay: float = 0.0,
        g_square_momentum: float = 1.0,
        eps: float = 1e-10,
        vectorwise_shared: bool = False,
    ):
        self.lr = lr
        self.initial_accumulator_value = initial_accumulator_value
        self.weight_decay = weight_decay  # accepted, unused by reference impl
        self.g_square_momentum = g_square_momentum
        self.eps = eps
        self.vectorwise_shared = vectorwise_shared

    def require_space(self, dim: int) -> int:
        return 1 if self.vectorwise_shared else dim

    def state_init(self, dim: int) -> float:
        return self.initial_accumulator_value
