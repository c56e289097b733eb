"""End-to-end adult-income-style training (mirrors the reference example
examples/src/adult-income/train.py, including its deterministic-mode AUC
reproducibility gate: with REPRODUCIBLE=1, EMBEDDING_STALENESS=1 and a fixed
seed, the final train AUC must reproduce bitwise run-to-run)."""
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
sys.path.insert(0, os.path.dirname(__file__))

from persia_amd.ctx import TrainCtx, eval_ctx
from persia_amd.data import DataLoader, IterableDataset
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.data import PersiaBatch
from persia_amd.embedding.optim import Adagrad
from persia_amd.logger import get_default_logger
from persia_amd.utils import setup_seed

from data_generator import make_dataloader, make_dataset
from model import DNN

logger = get_default_logger("adult_income")

REPRODUCIBLE = os.environ.get("REPRODUCIBLE", "0") in ("1", "true")
EMBEDDING_STALENESS = int(os.environ.get("EMBEDDING_STALENESS", "8"))
CONFIG_DIR = os.path.join(os.path.dirname(__file__), "config")


class TrainDataset(IterableDataset):
    def __init__(self, dense, ids, labels, batch_size=128, epochs=1):
        super().__init__(buffer_size=10)
        self.data = (dense, ids, labels)
        self.batch_size = batch_size
        self.epochs = epochs

    def __iter__(self):
        for _ in range(self.epochs):
            for dense, feats, label in make_dataloader(*self.data, self.batch_size):
                yield PersiaBatch(
                    feats, non_id_type_features=[dense], labels=[label], requires_grad=True
                )


def auc_score(labels: np.ndarray, preds: np.ndarray) -> float:
    order = np.argsort(preds, kind="mergesort")
    ranks = np.empty(len(preds))
    ranks[order] = np.arange(1, len(preds) + 1)
    pos = labels > 0.5
    n_pos, n_neg = pos.sum(), (~pos).sum()
    if n_pos == 0 or n_neg == 0:
        return 0.5
    return (ranks[pos].sum() - n_pos * (n_pos + 1) / 2) / (n_pos * n_neg)


def main(epochs=3, batch_size=128):
    if REPRODUCIBLE:
        setup_seed(3)
    dense, ids, labels = make_dataset()
    n_test = len(labels) // 5
    test_data = (dense[:n_test], ids[:n_test], labels[:n_test])
    train_data = (dense[n_test:], ids[n_test:], labels[n_test:])

    model = DNN()
    dense_opt = torch.optim.SGD(model.parameters(), lr=0.05)
    with TrainCtx(
        model=model,
        embedding_optimizer=Adagrad(lr=0.05),
        dense_optimizer=dense_opt,
        embedding_config=EmbeddingConfig(),
        embedding_schema=os.path.join(CONFIG_DIR, "embedding_config.yml"),
        global_config=os.path.join(CONFIG_DIR, "global_config.yml"),
        mixed_precision=False,
    ) as ctx:
        loader = DataLoader(
            TrainDataset(*train_data, batch_size=batch_size, epochs=epochs),
            reproducible=REPRODUCIBLE,
            embedding_staleness=1 if REPRODUCIBLE else EMBEDDING_STALENESS,
        )
        loss_fn = torch.nn.BCELoss()
        all_pred, all_label = [], []
        for i, batch in enumerate(loader):
            pred, lbls = ctx.forward(batch)
            loss = loss_fn(pred.squeeze(1), lbls[0].squeeze(1).float())
            ctx.backward(loss)
            all_pred.append(pred.detach().cpu().numpy())
            all_label.append(lbls[0].detach().cpu().numpy())
            if i % 50 == 0:
                logger.info(f"batch {i} loss {float(loss):.4f}")
        train_auc = auc_score(
            np.concatenate(all_label).ravel(), np.concatenate(all_pred).ravel()
        )
        logger.info(f"train auc: {train_auc!r}")

        # eval
        model.eval()
        with eval_ctx(model=model) as ectx:
            preds, lbl = [], []
            for d, feats, lab in make_dataloader(*test_data, batch_size):
                tb = ectx.engine.process_batch(
                    PersiaBatch(feats, non_id_type_features=[d], labels=[lab],
                                requires_grad=False)
                )
                p, l = ectx.forward(tb)
                preds.append(p.detach().cpu().numpy())
                lbl.append(l[0].cpu().numpy())
            test_auc = auc_score(np.concatenate(lbl).ravel(), np.concatenate(preds).ravel())
            logger.info(f"test auc: {test_auc!r}")
    return train_auc, test_auc


def train_and_dump(dst_dir, epochs=1, batch_size=128, online_batches=60):
    """Serving-path driver (the reference's train half of the TorchServe e2e,
    .buildkite/e2e docker-compose.train + serve_client.py): train, dump a
    full checkpoint, then KEEP training with incremental updates enabled so
    the infer side can prove it applies the freshness stream on top of the
    checkpoint.  Returns the incremental dir."""
    import itertools

    if REPRODUCIBLE:
        setup_seed(3)
    dense, ids, labels = make_dataset()
    n_test = len(labels) // 5
    train_data = (dense[n_test:], ids[n_test:], labels[n_test:])

    model = DNN()
    dense_opt = torch.optim.SGD(model.parameters(), lr=0.05)
    inc_dir = os.path.join(dst_dir, "inc")
    with TrainCtx(
        model=model,
        embedding_optimizer=Adagrad(lr=0.05),
        dense_optimizer=dense_opt,
        embedding_config=EmbeddingConfig(),
        embedding_schema=os.path.join(CONFIG_DIR, "embedding_config.yml"),
        global_config=os.path.join(CONFIG_DIR, "global_config.yml"),
        mixed_precision=False,
    ) as ctx:
        loss_fn = torch.nn.BCELoss()

        def run_epochs(n, limit=None):
            loader = DataLoader(
                TrainDataset(*train_data, batch_size=batch_size, epochs=n),
                reproducible=REPRODUCIBLE,
                embedding_staleness=1 if REPRODUCIBLE else EMBEDDING_STALENESS,
            )
            it = iter(loader)
            if limit is not None:
                it = itertools.islice(it, limit)
            for batch in it:
                pred, lbls = ctx.forward(batch)
                loss = loss_fn(pred.squeeze(1), lbls[0].squeeze(1).float())
                ctx.backward(loss)

        run_epochs(epochs)
        ctx.dump_checkpoint(dst_dir)
        # online phase: updates stream out as incremental packets
        inc = ctx.engine.enable_incremental_update(inc_dir, buffer_size=2000)
        run_epochs(1, limit=online_batches)
        inc.flush()
    return inc_dir


if __name__ == "__main__":
    train_auc, test_auc = main()
    print(f"train_auc={train_auc!r} test_auc={test_auc!r}")
