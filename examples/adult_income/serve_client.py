"""Serving-path evaluation (mirrors the reference
examples/src/adult-income/serve_client.py:77-79 infer-AUC gate, without
TorchServe/gRPC — the handler is driven in-process, which is the offline
analog of its gRPC round-trip: batches travel as serialized bytes through
``PersiaBatch.to_bytes`` -> ``InferCtx.get_embedding_from_bytes``).

Flow: boot a DNN + PersiaHandler from a dumped checkpoint, apply any pending
incremental-update packets, then score the held-out split through the full
serving path and print ``INFER_AUC <repr>``.
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
sys.path.insert(0, os.path.dirname(__file__))

from persia_amd.core.incremental import IncrementalUpdateLoader
from persia_amd.embedding.data import PersiaBatch

from data_generator import make_dataloader, make_dataset
from model import DNN
from serve_handler import PersiaHandler
from train import CONFIG_DIR, auc_score


def main(ckpt_dir: str, batch_size: int = 128) -> float:
    dense, ids, labels = make_dataset()
    n_test = len(labels) // 5
    test_data = (dense[:n_test], ids[:n_test], labels[:n_test])

    model = DNN()
    handler = PersiaHandler(
        model,
        embedding_schema=os.path.join(CONFIG_DIR, "embedding_config.yml"),
        checkpoint_dir=ckpt_dir,
    )
    handler.ctx.load_torch_state_dict(
        model, os.path.join(ckpt_dir, "dense.pt")
    )
    inc_dir = os.path.join(ckpt_dir, "inc")
    applied = 0
    if os.path.isdir(inc_dir):
        applied = IncrementalUpdateLoader(handler.ctx.engine, inc_dir).scan_once()
    preds, lbls = [], []
    for d, feats, lab in make_dataloader(*test_data, batch_size):
        batch = PersiaBatch(
            feats, non_id_type_features=[d], labels=[lab], requires_grad=False
        )
        tb = handler.preprocess(batch.to_bytes())  # the serving wire path
        preds.append(handler.inference(tb).numpy())
        lbls.append(tb.label_tensors[0].numpy())
    infer_auc = auc_score(
        np.concatenate(lbls).ravel(), np.concatenate(preds).ravel()
    )
    print(f"INFER_APPLIED {applied}")
    print(f"INFER_AUC {infer_auc!r}")
    return infer_auc


if __name__ == "__main__":
    main(sys.argv[1])
