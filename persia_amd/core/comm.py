"""Distributed communication for the sparse path.

The reference moves ids/rows/grads between processes over NATS + HTTP RPC
(persia-rpc, persia-nats-client).  On one MI355X node every shard owner is a
GPU rank, xGMI is fully connected (7 p2p links/GPU), and the natural
primitive is RCCL ``all_to_all_single`` with per-rank splits: each pair of
GPUs has a dedicated link, so the variable-length sign/row/grad exchange
saturates the mesh without a broker.

Backend "nccl" (= RCCL on ROCm) on GPU; "gloo" for CPU tests (world_size>1
multi-process CPU tests run in CI — gloo supports all_to_all_single with
uneven splits).
"""
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist


class DistContext:
    """Holds the process group used by the embedding engine (may be shared
    with DDP's default group)."""

    def __init__(self, world_size: int = 1, rank: int = 0, group=None):
        self.world_size = world_size
        self.rank = rank
        self.group = group

    @staticmethod
    def from_default_group() -> "DistContext":
        if dist.is_available() and dist.is_initialized():
            return DistContext(dist.get_world_size(), dist.get_rank(), None)
        return DistContext(1, 0, None)

    @staticmethod
    def new_sparse_group() -> "DistContext":
        """Dedicated communicator for the embedding all-to-all.

        The sparse exchange is issued from the lookup pipeline THREAD while
        DDP's all-reduce runs on the main thread; on one shared NCCL
        communicator their cross-rank issue order is not guaranteed and can
        deadlock.  A separate process group gives each side its own
        independently-ordered communicator (collectives within each are
        issued in the same order on every rank).  Must be called by ALL
        ranks."""
        if dist.is_available() and dist.is_initialized():
            group = dist.new_group(list(range(dist.get_world_size())))
            return DistContext(dist.get_world_size(), dist.get_rank(), group)
        return DistContext(1, 0, None)

    @property
    def distributed(self) -> bool:
        return self.world_size > 1

    def all_to_all_lengths(self, send_counts: List[int],
                           device: Optional[torch.device] = None) -> List[int]:
        """Exchange per-destination element counts (the two-phase counts/payload
        exchange the reference's RPC shape implies — SURVEY §7 hard part 3)."""
        inp = torch.tensor(send_counts, dtype=torch.int64)
        out = torch.empty_like(inp)
        if self._gloo_like():
            dist.all_to_all_single(out, inp, group=self.group)
        else:
            dev = device or torch.device("cuda", torch.cuda.current_device())
            inp_d = inp.to(dev, non_blocking=True)
            out_d = torch.empty_like(inp_d)
            dist.all_to_all_single(out_d, inp_d, group=self.group)
            out = out_d.cpu()
        return out.tolist()

    def _gloo_like(self) -> bool:
        backend = dist.get_backend(self.group)
        return "gloo" in str(backend)

    def all_to_all(
        self,
        inp: torch.Tensor,
        send_counts: List[int],
        recv_counts: List[int],
    ) -> torch.Tensor:
        """Variable all-to-all of a flat (or 2D row-major) tensor.

        ``send_counts``/``recv_counts`` are in UNITS OF ROWS of ``inp``."""
        out_rows = sum(recv_counts)
        shape = (out_rows,) + tuple(inp.shape[1:])
        out = torch.empty(shape, dtype=inp.dtype, device=inp.device)
        dist.all_to_all_single(
            out, inp.contiguous(), recv_counts, send_counts, group=self.group
        )
        return out

    def all_to_all_even(self, inp: torch.Tensor,
                        out: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Fixed-shape all-to-all: every rank sends/receives exactly
        ``inp.shape[0] // world_size`` rows per peer.  This is the sync-free
        exchange of the fused distributed path (capacity-padded buckets; the
        variable per-destination counts never reach the host).  ``out`` may
        be a contiguous view (e.g. the prefix of a buffer with a dummy row)."""
        if out is None:
            out = torch.empty_like(inp)
        dist.all_to_all_single(out, inp.contiguous(), group=self.group)
        return out

    def barrier(self):
        if self.distributed:
            dist.barrier(group=self.group)

    def allreduce_scalar(self, x: float) -> float:
        if not self.distributed:
            return x
        t = torch.tensor([x], dtype=torch.float64)
        if not self._gloo_like():
            # RCCL rejects CPU tensors: stage through the device like
            # all_to_all_lengths does
            dev = torch.device("cuda", torch.cuda.current_device())
            t_d = t.to(dev)
            dist.all_reduce(t_d, group=self.group)
            t = t_d.cpu()
        else:
            dist.all_reduce(t, group=self.group)
        return float(t.item())

    def allreduce_int_minmax(self, x: int) -> Tuple[int, int]:
        """(min, max) of an int across ranks — one-time shape checks for the
        padded a2a fast path (all ranks must agree on nnz/cap)."""
        if not self.distributed:
            return x, x
        t = torch.tensor([x, -x], dtype=torch.int64)
        if not self._gloo_like():
            dev = torch.device("cuda", torch.cuda.current_device())
            t_d = t.to(dev)
            dist.all_reduce(t_d, op=dist.ReduceOp.MAX, group=self.group)
            t = t_d.cpu()
        else:
            dist.all_reduce(t, op=dist.ReduceOp.MAX, group=self.group)
        return -int(t[1].item()), int(t[0].item())
