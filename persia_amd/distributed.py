"""Distributed options for the dense (synchronous) side.

Mirrors reference ``persia/distributed.py``: ``DDPOption`` configures
``torch.distributed`` + ``DistributedDataParallel``.  On ROCm the "nccl"
backend IS RCCL over xGMI.  The reference's Bagua option (persia/distributed.py:204-256)
has no ROCm build; :class:`BaguaDistributedOption` is kept for API parity and
maps its algorithms onto DDP equivalents where possible.
"""
import os
from abc import ABC
from typing import Optional

import torch
import torch.distributed as dist

from persia_amd.env import get_master_addr, get_master_port, get_rank, get_world_size
from persia_amd.logger import get_default_logger

_logger = get_default_logger("persia_amd.distributed")


class DistributedBaseOption(ABC):
    def __init__(self, master_addr: Optional[str] = None, master_port: Optional[int] = None):
        self.master_addr = master_addr
        self.master_port = master_port

    def init_process_group(self, device_id: Optional[int]) -> None:
        raise NotImplementedError

    def wrap_model(self, model: torch.nn.Module, device_id: Optional[int]) -> torch.nn.Module:
        raise NotImplementedError

    def post_optimizer_step(self, model: torch.nn.Module) -> None:
        """Called by TrainCtx.backward after each dense optimizer step
        (no-op for synchronous algorithms)."""


class DDPOption(DistributedBaseOption):
    """torch DDP over RCCL (reference persia/distributed.py:74-193).

    ``backend`` defaults to "nccl" (RCCL) on GPU, "gloo" on CPU.
    """

    def __init__(
        self,
        backend: Optional[str] = None,
        master_addr: Optional[str] = None,
        master_port: Optional[int] = None,
        find_unused_parameters: bool = True,
        gradient_as_bucket_view: bool = True,
        bucket_cap_mb: int = 50,
        **options,
    ):
        super().__init__(master_addr, master_port)
        self.backend = backend
        self.find_unused_parameters = find_unused_parameters
        self.gradient_as_bucket_view = gradient_as_bucket_view
        # xGMI is per-link bound (7 × ~153 GB/s): bigger buckets amortize the
        # per-collective latency; DDP default 25 MB is tuned for NVLink.
        self.bucket_cap_mb = bucket_cap_mb
        self.options = options

    def resolved_backend(self, device_id: Optional[int]) -> str:
        if self.backend:
            return self.backend
        return "nccl" if device_id is not None else "gloo"

    def init_process_group(self, device_id: Optional[int]) -> None:
        if dist.is_initialized():
            return
        backend = self.resolved_backend(device_id)
        addr = self.master_addr or get_master_addr()
        port = self.master_port or get_master_port()
        os.environ.setdefault("MASTER_ADDR", addr)
        os.environ.setdefault("MASTER_PORT", str(port))
        dist.init_process_group(
            backend=backend,
            rank=get_rank(),
            world_size=get_world_size(),
        )
        _logger.info(
            f"initialized process group backend={backend} rank={get_rank()} "
            f"world_size={get_world_size()}"
        )

    def wrap_model(self, model: torch.nn.Module, device_id: Optional[int]) -> torch.nn.Module:
        from torch.nn.parallel import DistributedDataParallel

        kwargs = dict(
            find_unused_parameters=self.find_unused_parameters,
            gradient_as_bucket_view=self.gradient_as_bucket_view,
            bucket_cap_mb=self.bucket_cap_mb,
        )
        if device_id is not None:
            kwargs["device_ids"] = [device_id]
        return DistributedDataParallel(model, **kwargs)


class BaguaDistributedOption(DistributedBaseOption):
    """The reference's Bagua option (persia/distributed.py:204-256) mapped to
    RCCL-native equivalents (Bagua itself has no ROCm build):

    * ``gradient_allreduce``      -> plain DDP (identical algorithm)
    * ``bytegrad`` / ``qadam``    -> DDP + bf16-compressed allreduce hook
      (torch's powerSGD-family comm hooks over RCCL; Bagua's 8-bit codec is
      approximated by bf16 compression — half the wire bytes)
    * ``low_precision_decentralized`` -> same compressed hook (decentralized
      topologies are pointless on a fully-connected xGMI node)
    * ``decentralized``           -> DDP with a wider bucket (peer averaging
      on an 8-GPU all-to-all mesh degenerates to allreduce)
    * ``async``                   -> periodic (local-SGD style) model
      averaging: NO per-step gradient sync — each rank trains locally and
      every ``sync_every_steps`` optimizer steps the weights are averaged
      with one allreduce (the collective-safe mapping of Bagua's async
      model-average algorithm — its lock-free peer pulls cannot be expressed
      with collectives, but the statistical behaviour, bounded weight drift
      between periodic merges, is the same).  ``sync_interval_ms`` is
      accepted for reference API parity and treated as a step interval.
    """

    _COMPRESSED = {"bytegrad", "qadam", "low_precision_decentralized"}
    _SUPPORTED = _COMPRESSED | {"gradient_allreduce", "decentralized", "async"}

    def __init__(self, algorithm: str = "gradient_allreduce", **options):
        super().__init__()
        self.algorithm = algorithm
        if algorithm not in self._SUPPORTED:
            raise NotImplementedError(
                f"Bagua algorithm {algorithm!r} has no MI355X mapping "
                f"(supported: {sorted(self._SUPPORTED)})"
            )
        self.sync_every_steps = int(
            options.pop("sync_every_steps",
                        options.pop("sync_interval_ms", 8) or 8)
        )
        self._step = 0
        bucket = 100 if algorithm == "decentralized" else 50
        self._ddp = DDPOption(bucket_cap_mb=bucket, **options)

    def init_process_group(self, device_id):
        self._ddp.init_process_group(device_id)

    def wrap_model(self, model, device_id):
        if self.algorithm == "async":
            # local training: no DDP wrap, but start from identical weights
            with torch.no_grad():
                for t in list(model.parameters()) + list(model.buffers()):
                    dist.broadcast(t.data, src=0)
            return model
        wrapped = self._ddp.wrap_model(model, device_id)
        if self.algorithm in self._COMPRESSED:
            from torch.distributed.algorithms.ddp_comm_hooks import (
                default_hooks,
            )

            wrapped.register_comm_hook(
                state=None, hook=default_hooks.bf16_compress_hook
            )
        return wrapped

    def post_optimizer_step(self, model):
        if self.algorithm != "async" or not dist.is_initialized():
            return
        self._step += 1
        if self._step % self.sync_every_steps:
            return
        world = dist.get_world_size()
        with torch.no_grad():
            for p in model.parameters():
                dist.all_reduce(p.data)
                p.data.div_(world)


def get_default_distributed_option(device_id: Optional[int] = None) -> DDPOption:
    """Reference persia/distributed.py:413-428: pick by device."""
    return DDPOption(backend="nccl" if device_id is not None else "gloo")
