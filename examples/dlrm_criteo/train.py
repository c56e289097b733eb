"""DLRM-Criteo-shape training through the PUBLIC API (TrainCtx + DataLoader
+ PersiaBatch) — the user-journey twin of the flagship bench.py (which
drives the engine directly for the graphed hot loop).  Synthetic uniform
IDs / random labels, random-init weights (no network access), CPU-safe at
small sizes; on MI355X pass --fused for the MFMA dense layers.

Usage:
  python examples/dlrm_criteo/train.py [--steps 50] [--batch-size 512]
      [--num-sparse 26] [--dim 32] [--rows 1e5] [--fused]

Reference analog: PERSIA's DLRM deployments drive the same ctx/data API
(reference persia/ctx.py:753-1064 + persia/data.py:202-271).
"""
import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig
from persia_amd.ctx import TrainCtx
from persia_amd.data import DataLoader, IterableDataset
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.data import (
    IDTypeFeatureWithSingleID,
    Label,
    NonIDTypeFeature,
    PersiaBatch,
)
from persia_amd.embedding.optim import Adagrad
from persia_amd.logger import get_default_logger
from persia_amd.models import DLRM

logger = get_default_logger("dlrm_criteo")


class SyntheticCriteo(IterableDataset):
    def __init__(self, steps, batch_size, num_sparse, num_dense, rows, seed=0):
        super().__init__(buffer_size=10)
        self.steps = steps
        self.B = batch_size
        self.num_sparse = num_sparse
        self.num_dense = num_dense
        self.rows = rows
        self.seed = seed

    def __iter__(self):
        rng = np.random.default_rng(self.seed)
        vocab = max(1, self.rows // self.num_sparse)
        for _ in range(self.steps):
            feats = [
                IDTypeFeatureWithSingleID(
                    f"f{i}",
                    rng.integers(0, vocab, size=self.B, dtype=np.uint64),
                )
                for i in range(self.num_sparse)
            ]
            dense = rng.random((self.B, self.num_dense), dtype=np.float32)
            # learnable synthetic target mixing a dense signal (fast, via
            # the bottom MLP) with a sparse one (slot-0 id parity, reachable
            # only through that slot's embedding and the dot interaction):
            # loss must fall below ln(2) if the hybrid path actually trains
            score = dense[:, 0] + 0.25 * (feats[0].data % 2)
            label = (score > 0.62).astype(np.float32).reshape(-1, 1)
            yield PersiaBatch(
                feats,
                non_id_type_features=[NonIDTypeFeature(dense)],
                labels=[Label(label)],
                requires_grad=True,
            )


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--batch-size", type=int, default=512)
    p.add_argument("--num-sparse", type=int, default=26)
    p.add_argument("--num-dense", type=int, default=13)
    p.add_argument("--dim", type=int, default=32)
    p.add_argument("--rows", type=float, default=1e5)
    p.add_argument("--fused", action="store_true",
                   help="hand-written MFMA dense layers (GPU)")
    args = p.parse_args(argv)

    use_gpu = torch.cuda.is_available()
    schema = EmbeddingSchema(
        slots={
            f"f{i}": SlotConfig(name=f"f{i}", dim=args.dim)
            for i in range(args.num_sparse)
        }
    )
    model = DLRM(
        num_sparse=args.num_sparse, num_dense=args.num_dense, dim=args.dim,
        fused=args.fused and use_gpu,
    )
    dense_opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    losses = []
    with TrainCtx(
        model=model,
        embedding_optimizer=Adagrad(lr=0.02),
        dense_optimizer=dense_opt,
        embedding_config=EmbeddingConfig(),
        embedding_schema=schema,
        global_config=GlobalConfig(capacity=1 << 20),
        mixed_precision=use_gpu,
    ) as ctx:
        loader = DataLoader(
            SyntheticCriteo(args.steps, args.batch_size, args.num_sparse,
                            args.num_dense, int(args.rows)),
        )
        loss_fn = torch.nn.BCEWithLogitsLoss()
        for step, tb in enumerate(loader):
            dense, embs, labels = ctx.prepare_features(tb)
            logits = ctx.model(dense, embs)
            loss = loss_fn(logits.float(), labels[0].squeeze(-1))
            ctx.backward(loss)
            losses.append(float(loss.detach()))
            if step % 10 == 0:
                logger.info(f"step {step}: loss {losses[-1]:.4f}")
    first = np.mean(losses[: max(1, len(losses) // 5)])
    last = np.mean(losses[-max(1, len(losses) // 5):])
    print(f"LOSS_FIRST {first:.5f}")
    print(f"LOSS_LAST {last:.5f}")
    return first, last


if __name__ == "__main__":
    main()
