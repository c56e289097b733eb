"""Model-serving server: the runnable analog of the reference's TorchServe
integration.

The reference serves through TorchServe + gRPC
(resources/proto/inference.proto: ``Ping`` + ``Predictions(map<string,
bytes>) -> bytes``; examples/src/adult-income/serve_client.py drives it).
TorchServe is absent offline, so this module provides the same surface as a
self-contained HTTP service over the byte wire:

* ``GET  /ping``                      -> ``{"health": "healthy"}``
* ``POST /predictions/{model_name}``  -> raw little-endian f32 predictions
  (request body = ``PersiaBatch.to_bytes`` wire bytes, exactly what the
  reference puts in the gRPC ``input`` map)

:class:`PersiaHandler` is the TorchServe-custom-handler-shaped core
(preprocess bytes -> InferCtx lookup -> dense forward) and is usable
directly in-process; :func:`create_app` wraps it for HTTP, :func:`serve`
runs uvicorn.  Online freshness comes from attaching an
:class:`~persia_amd.core.incremental.IncrementalUpdateLoader`.
"""
from typing import List, Optional

import torch

from persia_amd.ctx import InferCtx
from persia_amd.logger import get_default_logger

_logger = get_default_logger("persia_amd.serving")


class PersiaHandler:
    """TorchServe-custom-handler shape: preprocess -> inference -> handle."""

    def __init__(self, model: torch.nn.Module, embedding_schema,
                 checkpoint_dir: Optional[str] = None,
                 incremental_dir: Optional[str] = None,
                 poll_interval_sec: float = 10.0):
        self.model = model
        self.model.eval()
        self.ctx = InferCtx(embedding_schema=embedding_schema, model=model)
        if checkpoint_dir:
            self.ctx.load_embedding(checkpoint_dir)
        self.incremental = None
        if incremental_dir:
            from persia_amd.core.incremental import IncrementalUpdateLoader

            self.incremental = IncrementalUpdateLoader(
                self.ctx.engine, incremental_dir,
                poll_interval_sec=poll_interval_sec,
            )
            self.incremental.scan_once()
            self.incremental.start()

    def preprocess(self, data: bytes):
        return self.ctx.get_embedding_from_bytes(data)

    def inference(self, batch) -> torch.Tensor:
        with torch.no_grad():
            pred, _labels = self.ctx.forward(batch)
        return pred

    def handle(self, data: bytes) -> List[float]:
        return self.inference(self.preprocess(data)).view(-1).tolist()


def create_app(handler: PersiaHandler, model_name: str = "persia"):
    import numpy as np
    from fastapi import FastAPI, HTTPException, Request, Response

    app = FastAPI(title="persia-inference")

    @app.get("/ping")
    def ping():
        return {"health": "healthy"}

    @app.post("/predictions/{name}")
    async def predictions(name: str, request: Request):
        if name != model_name:
            raise HTTPException(404, f"unknown model {name!r}")
        body = await request.body()
        try:
            preds = handler.handle(body)
        except Exception as e:
            raise HTTPException(400, f"bad request: {e}")
        return Response(
            content=np.asarray(preds, dtype="<f4").tobytes(),
            media_type="application/octet-stream",
        )

    return app


def serve(handler: PersiaHandler, port: int, model_name: str = "persia"):
    import uvicorn

    _logger.info(f"inference server on :{port} model={model_name}")
    uvicorn.run(create_app(handler, model_name), host="0.0.0.0", port=port)
