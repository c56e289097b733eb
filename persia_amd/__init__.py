"""persia_amd — an MI355X-native hybrid-parallel recommender training framework.

A from-scratch rebuild of the capabilities of PersiaML/PERSIA (reference:
``persia`` Python package + Rust service tier) designed for a single node of
AMD Instinct MI355X GPUs:

* trillion-scale embedding tables live **in HBM3E** as sharded GPU hash
  tables (one shard per GPU, optional host-DRAM spill), not on remote CPU
  parameter servers;
* the lookup / gradient hot path is HIP kernels + RCCL ``all_to_all_single``
  over xGMI, not HTTP RPC;
* the dense model trains synchronously with DDP over RCCL while the sparse
  side updates asynchronously with bounded staleness — PERSIA's hybrid
  algorithm (reference: ``rust/persia-core/src/forward.rs``).

Public API mirrors the reference ``persia`` package: ``persia_amd.ctx``,
``persia_amd.data``, ``persia_amd.embedding``, ``persia_amd.distributed``,
``persia_amd.env``, ``persia_amd.logger``, ``persia_amd.utils``.
"""

from persia_amd.version import __version__

__all__ = ["__version__"]
