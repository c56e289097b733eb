"""Inference handler (mirrors the reference TorchServe integration,
examples/src/adult-income/serve_handler.py: preprocess bytes -> InferCtx
embedding lookup -> dense model forward).

Works standalone or as a TorchServe custom handler."""
import os
import sys
from typing import List

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from persia_amd.ctx import InferCtx
from persia_amd.service import get_embedding_worker_services


class PersiaHandler:
    def __init__(self, model: torch.nn.Module, embedding_schema, checkpoint_dir=None):
        self.model = model
        self.model.eval()
        self.ctx = InferCtx(
            embedding_schema=embedding_schema,
            model=model,
        )
        if checkpoint_dir:
            self.ctx.load_embedding(checkpoint_dir)

    def preprocess(self, data: bytes):
        return self.ctx.get_embedding_from_bytes(data)

    def inference(self, batch) -> torch.Tensor:
        with torch.no_grad():
            pred, _labels = self.ctx.forward(batch)
        return pred

    def handle(self, data: bytes) -> List[float]:
        return self.inference(self.preprocess(data)).view(-1).tolist()
