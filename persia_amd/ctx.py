"""Context managers defining the process roles (mirrors reference persia/ctx.py).

* :class:`BaseCtx`     — engine + rank bootstrap (ctx.py:202-272)
* :class:`DataCtx`     — batch producer role (ctx.py:274-343)
* :class:`EmbeddingCtx`— forward / feature preparation / checkpoints (ctx.py:345-652)
* :class:`TrainCtx`    — + backward, AMP, DDP (ctx.py:655-1064)
* :class:`InferCtx`    — inference without NATS (ctx.py:1077-1133)
* :func:`eval_ctx`     — EmbeddingCtx in EVAL mode (ctx.py:1072)

Unlike the reference, there are no remote embedding workers or parameter
servers: every ctx drives the in-process :class:`EmbeddingEngine`, whose
table shard lives in this rank's HBM.
"""
import os
from enum import Enum
from queue import Queue
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from persia_amd import env as _env
from persia_amd.core.comm import DistContext
from persia_amd.core.engine import EmbeddingEngine, PersiaTrainingBatch
from persia_amd.core.schema import EmbeddingSchema, GlobalConfig
from persia_amd.distributed import DDPOption, DistributedBaseOption, get_default_distributed_option
from persia_amd.embedding import EmbeddingConfig
from persia_amd.embedding.data import PersiaBatch
from persia_amd.embedding.optim import Optimizer
from persia_amd.logger import get_default_logger

_logger = get_default_logger("persia_amd.ctx")

_CURRENT_CTX = None
_LAST_ENGINE: Optional[EmbeddingEngine] = None


def _check_finite(tensors: List[torch.Tensor]) -> bool:
    """reference ctx.py:28-37"""
    return all([torch.isfinite(t).all() if t is not None else True for t in tensors])


class PreprocessMode(Enum):
    """reference ctx.py:58-73"""

    TRAIN = 1
    EVAL = 2
    INFERENCE = 3


def _load_schema(embedding_schema) -> EmbeddingSchema:
    if isinstance(embedding_schema, EmbeddingSchema):
        return embedding_schema
    if isinstance(embedding_schema, dict):
        return EmbeddingSchema.from_dict(embedding_schema)
    if isinstance(embedding_schema, str):
        return EmbeddingSchema.from_yaml(embedding_schema)
    path = os.environ.get("PERSIA_EMBEDDING_CONFIG")
    if path:
        return EmbeddingSchema.from_yaml(path)
    raise ValueError(
        "no embedding schema: pass embedding_schema= (EmbeddingSchema, dict or "
        "yaml path) or set PERSIA_EMBEDDING_CONFIG"
    )


def _load_gconf(global_config) -> GlobalConfig:
    if isinstance(global_config, GlobalConfig):
        return global_config
    if isinstance(global_config, str):
        return GlobalConfig.from_yaml(global_config)
    path = os.environ.get("PERSIA_GLOBAL_CONFIG")
    if path:
        return GlobalConfig.from_yaml(path)
    return GlobalConfig()


class BaseCtx:
    """Bootstraps rank info and (for embedding-carrying ctxs) the engine."""

    def __init__(self, threadpool_worker_size: int = 10, device_id: Optional[int] = None):
        from persia_amd.core.watchdog import maybe_start_deadlock_detection

        maybe_start_deadlock_detection()
        if device_id is None:
            device_id = _env.get_local_rank() if torch.cuda.is_available() else None
        if device_id is not None and device_id >= 0:
            assert torch.cuda.is_available(), "device_id set but no GPU available"
            torch.cuda.set_device(device_id)
            self.device = torch.device("cuda", device_id)
            self.device_id: Optional[int] = device_id
        else:
            self.device = torch.device("cpu")
            self.device_id = None
        self.rank = _env.get_rank()
        self.world_size = _env.get_world_size()

    def _enter(self):
        global _CURRENT_CTX
        self._prev_ctx = _CURRENT_CTX
        _CURRENT_CTX = self

    def _exit(self):
        global _CURRENT_CTX
        _CURRENT_CTX = self._prev_ctx

    def __enter__(self):
        self._enter()
        return self

    def __exit__(self, exc_type, value, trace):
        self._exit()
        if exc_type is not None:
            import traceback

            _logger.error("\n" + "".join(traceback.format_tb(trace)))


class DataCtx(BaseCtx):
    """Producer role: sends :class:`PersiaBatch` into the dataflow
    (reference ctx.py:274-343 — NATS publish; here an in-process channel or a
    TCP queue to the trainer, persia_amd/core/queue.py)."""

    def __init__(self, sink=None, **kwargs):
        super().__init__(**kwargs)
        from persia_amd.core.queue import get_default_sink

        self.sink = sink or get_default_sink()
        self._sent = 0

    def send_data(self, persia_batch: PersiaBatch, block: bool = True):
        persia_batch.batch_id = (
            self._sent * _env.get_replica_size() + _env.get_replica_index()
        )
        self._sent += 1
        self.sink.send(persia_batch, block=block)


class EmbeddingCtx(BaseCtx):
    def __init__(
        self,
        preprocess_mode: PreprocessMode = PreprocessMode.EVAL,
        model: Optional[torch.nn.Module] = None,
        embedding_config: Optional[EmbeddingConfig] = None,
        embedding_schema=None,
        embedding_optimizer: Optional[Optimizer] = None,
        global_config=None,
        engine: Optional[EmbeddingEngine] = None,
        **kwargs,
    ):
        super().__init__(**kwargs)
        self.preprocess_mode = preprocess_mode
        self.model = model
        if model is not None and self.device.type == "cuda":
            model.to(self.device)  # reference TrainCtx moves the dense model
        self.embedding_config = embedding_config or EmbeddingConfig()
        global _LAST_ENGINE
        if engine is not None:
            self.engine = engine
        elif (
            embedding_schema is None
            and global_config is None
            and _LAST_ENGINE is not None
        ):
            # reuse the process engine (reference: PersiaCommonContext is a
            # process singleton, persia-core/src/lib.rs:100-130)
            self.engine = _LAST_ENGINE
        else:
            schema = _load_schema(embedding_schema)
            gconf = _load_gconf(global_config)
            self.engine = EmbeddingEngine(
                schema=schema,
                hyper=self.embedding_config,
                optimizer=embedding_optimizer,
                gconf=gconf,
                device=self.device,
                # own communicator: the sparse all-to-all must not interleave
                # with DDP's all-reduce on one NCCL comm (deadlock hazard)
                dist_ctx=DistContext.new_sparse_group(),
            )
        _LAST_ENGINE = self.engine
        self.current_batch: Optional[PersiaTrainingBatch] = None

    # ------------------------------------------------------------- features

    def prepare_features(
        self, batch: PersiaTrainingBatch, mode: Optional[PreprocessMode] = None
    ) -> Tuple[List[torch.Tensor], List[torch.Tensor], Optional[List[torch.Tensor]]]:
        """Builds the model-input tensors; exact contract of reference
        _prepare_feature (ctx.py:75-199): sum slots -> (B, dim) f16 tensor;
        raw slots -> (B, sample_fixed_size, dim+1) f16 with a 0/1 mask as the
        last channel."""
        mode = mode or self.preprocess_mode
        self.current_batch = batch
        if mode == PreprocessMode.INFERENCE:
            labels = None
        else:
            labels = batch.label_tensors
        is_training = mode == PreprocessMode.TRAIN and batch.requires_grad

        # sum slots: gradients flow into the per-group BASE tensor (one fused
        # scatter in backward instead of a per-slot cascade)
        sum_views = batch.enable_training_views() if is_training else {}

        emb_tensors: List[torch.Tensor] = []
        cache = []  # (name, distinct_id_tensor, index, non_empty_index, emb_tensor)
        for p in batch.payloads:
            if p.is_raw:
                distinct_id_tensor = p.raw_distinct
                index_tensor = p.raw_index
                non_empty_index = p.raw_non_empty_index
                batch_size = batch.batch_size
                dim = distinct_id_tensor.shape[-1]
                sample_fixed_size = index_tensor.shape[-1] // batch_size
                index_select_raw_tensor = distinct_id_tensor.index_select(
                    0, index_tensor.view(-1)
                )
                index_select_raw_tensor.requires_grad = is_training
                raw_fixed_size_tensor = index_select_raw_tensor.view(
                    -1, sample_fixed_size, dim
                )
                mask = (
                    index_tensor.view(batch_size, sample_fixed_size, 1) != 0
                ).half()
                emb_tensors.append(torch.cat([raw_fixed_size_tensor, mask], dim=2))
                cache.append(
                    (p.name, distinct_id_tensor, index_tensor, non_empty_index, index_select_raw_tensor)
                )
            else:
                t = sum_views.get(p.name, p.sum_tensor)
                emb_tensors.append(t)
                cache.append((p.name, None, None, None, t))
        batch._emb_cache = cache
        return batch.non_id_type_tensors, emb_tensors, labels

    def forward(
        self, batch: PersiaTrainingBatch
    ) -> Tuple[torch.Tensor, Optional[List[torch.Tensor]]]:
        assert self.model is not None, "model not found, please init ctx with model"
        non_id, embs, labels = self.prepare_features(batch)
        output = self.model(non_id, embs)
        return output, labels

    # ----------------------------------------------------------- checkpoints

    def dump_checkpoint(
        self,
        dst_dir: str,
        dense_model_filename: str = "dense.pt",
        jit_dense_model_filename: str = "jit_dense.pt",
        blocking: bool = True,
        with_jit_model: bool = False,
    ) -> None:
        os.makedirs(dst_dir, exist_ok=True)
        if self.model is not None and self.engine.dist.rank == 0:
            self.dump_torch_state_dict(
                self.model, dst_dir, dense_model_filename, is_embedding_optimizer=False
            )
            if with_jit_model:
                jit = torch.jit.script(_unwrap_ddp(self.model))
                torch.jit.save(jit, os.path.join(dst_dir, jit_dense_model_filename))
        self.dump_embedding(dst_dir, blocking=blocking)

    def load_checkpoint(
        self,
        src_dir: str,
        map_location=None,
        dense_model_filename: str = "dense.pt",
        blocking: bool = True,
    ) -> None:
        dense_path = os.path.join(src_dir, dense_model_filename)
        if self.model is not None and os.path.exists(dense_path):
            self.load_torch_state_dict(self.model, dense_path, map_location)
        self.load_embedding(src_dir, blocking=blocking)

    def dump_embedding(self, dst_dir: str, blocking: bool = True) -> None:
        self.engine.dump(dst_dir, blocking=blocking)

    def load_embedding(self, src_dir: str, blocking: bool = True) -> None:
        self.engine.load(src_dir, blocking=blocking)

    def dump_torch_state_dict(
        self, torch_instance, dst_dir: str, filename: str, is_embedding_optimizer=False
    ) -> None:
        obj = _unwrap_ddp(torch_instance)
        state = obj.state_dict() if hasattr(obj, "state_dict") else obj
        torch.save(state, os.path.join(dst_dir, filename))

    def load_torch_state_dict(self, torch_instance, src_path: str, map_location=None) -> None:
        state = torch.load(src_path, map_location=map_location or "cpu")
        _unwrap_ddp(torch_instance).load_state_dict(state)

    def get_embedding_size(self) -> List[int]:
        """Resident rows per dim-group shard, summed over ranks."""
        sizes = [len(s) for _d, s in sorted(self.engine.stores.items())]
        if self.engine.dist.distributed:
            sizes = [int(self.engine.dist_grad.allreduce_scalar(float(s))) for s in sizes]
        return sizes

    def clear_embeddings(self) -> None:
        for s in self.engine.stores.values():
            s.clear()


def _unwrap_ddp(m):
    return m.module if hasattr(m, "module") and isinstance(
        m, torch.nn.parallel.DistributedDataParallel
    ) else m


class TrainCtx(EmbeddingCtx):
    """Training role (reference ctx.py:655-1064): synchronous dense DDP +
    asynchronous sparse updates with bounded staleness."""

    def __init__(
        self,
        model: Optional[torch.nn.Module] = None,
        embedding_optimizer: Optional[Optimizer] = None,
        dense_optimizer: Optional[torch.optim.Optimizer] = None,
        grad_scalar_update_factor: float = 4.0,
        backward_buffer_size: int = 10,
        backward_workers_size: int = 8,
        grad_update_buffer_size: int = 60,
        lookup_emb_directly: bool = True,
        mixed_precision: bool = True,
        distributed_option: Optional[DistributedBaseOption] = None,
        **kwargs,
    ):
        kwargs.setdefault("preprocess_mode", PreprocessMode.TRAIN)
        # distributed init must precede engine creation (the engine snapshots
        # the default process group)
        device_id = kwargs.get("device_id")
        if device_id is None and torch.cuda.is_available():
            device_id = _env.get_local_rank()
        self.distributed_option = distributed_option
        if _env.get_world_size() > 1:
            self.distributed_option = distributed_option or get_default_distributed_option(
                device_id if torch.cuda.is_available() else None
            )
            self.distributed_option.init_process_group(
                device_id if torch.cuda.is_available() else None
            )
        super().__init__(
            model=model, embedding_optimizer=embedding_optimizer, **kwargs
        )
        assert model is not None, "TrainCtx requires model"
        assert dense_optimizer is not None, "TrainCtx requires dense_optimizer"
        assert grad_scalar_update_factor > 0
        self.dense_optimizer = dense_optimizer
        self.grad_scalar_update_factor = grad_scalar_update_factor
        self.update_times = 0
        self.mixed_precision = mixed_precision and self.device.type == "cuda"
        if self.mixed_precision:
            self.grad_scaler = torch.cuda.amp.GradScaler()
        if self.world_size > 1 and self.distributed_option is not None:
            self.model = self.distributed_option.wrap_model(self.model, self.device_id)
        self.grad_queue: Queue = Queue(grad_update_buffer_size)
        self._pipeline = None  # attached by DataLoader

    def backward(
        self, loss: torch.Tensor, embedding_gradient_check_frequency: int = 20
    ) -> torch.Tensor:
        """reference ctx.py:893-924"""
        if self.mixed_precision:
            loss = self.grad_scaler.scale(loss)
            scale = self.grad_scaler.get_scale()
        else:
            scale = 1.0
        loss.backward()
        finite = self._on_backward(scale, embedding_gradient_check_frequency)
        if self.mixed_precision:
            self.grad_scaler.step(self.dense_optimizer)
            if finite:
                self.grad_scaler.update()
            else:
                self.grad_scaler.update(scale / self.grad_scalar_update_factor)
        else:
            self.dense_optimizer.step()
        self.dense_optimizer.zero_grad()
        if self.distributed_option is not None:
            # periodic model averaging for the Bagua-async mapping (no-op
            # for synchronous options)
            self.distributed_option.post_optimizer_step(self.model)
        return loss

    def _on_backward(self, loss_scale: float, check_frequency: int) -> bool:
        """reference ctx.py:926-1005 — builds per-slot gradient tensors and
        pushes them into the engine (raw slots: index_add_ de-dup scatter,
        dropping the padding row 0)."""
        batch = self.current_batch
        finite = True
        if self.mixed_precision and self.update_times % check_frequency == 0:
            finite = _check_finite(
                [g.sum_base.grad for g in batch._groups if g.sum_base is not None]
                + [
                    c[-1].grad
                    for c in batch._emb_cache
                    if c[1] is not None  # raw slots: the index-selected tensor
                ]
            )
        self.update_times += 1

        # raw slots: per-slot de-dup scatter (reference ctx.py:968-976);
        # sum slots: gradients are already in each group's sum_base.grad
        raw_grads: Dict[str, Optional[torch.Tensor]] = {}
        for (name, distinct_id_tensor, index, non_empty_index, emb_tensor) in batch._emb_cache:
            if distinct_id_tensor is None:
                continue
            if emb_tensor.grad is None:
                raw_grads[name] = None
                continue
            if distinct_id_tensor.shape[0] > 1:
                grad = torch.zeros_like(distinct_id_tensor, dtype=torch.float32)
                nz = non_empty_index.view(-1)
                non_zero_grad = emb_tensor.grad.index_select(0, nz).float()
                dst = index.view(-1)[nz]
                grad.index_add_(0, dst, non_zero_grad)
                raw_grads[name] = grad[1:, :]
            else:
                raw_grads[name] = None
        self.engine.apply_gradients_base(batch, raw_grads, loss_scale)
        if self._pipeline is not None and batch.requires_grad:
            self._pipeline.release_permit()
        return finite


class InferCtx(EmbeddingCtx):
    """Inference role (reference ctx.py:1077-1133): read-only lookups, zeros
    on miss."""

    def __init__(self, *args, **kwargs):
        kwargs.setdefault("preprocess_mode", PreprocessMode.INFERENCE)
        super().__init__(*args, **kwargs)
        self.engine.gconf.job_type = "infer"

    def get_embedding_from_data(self, batch: PersiaBatch):
        return self.engine.process_batch(batch, train=False)

    def get_embedding_from_bytes(self, data: bytes):
        return self.get_embedding_from_data(PersiaBatch.from_bytes(data))


def eval_ctx(*args, **kwargs) -> EmbeddingCtx:
    """reference ctx.py:1072-1075"""
    kwargs.setdefault("preprocess_mode", PreprocessMode.EVAL)
    if "engine" not in kwargs and _CURRENT_CTX is not None and hasattr(_CURRENT_CTX, "engine"):
        kwargs["engine"] = _CURRENT_CTX.engine
    return EmbeddingCtx(*args, **kwargs)


def cnt_ctx():
    return _CURRENT_CTX
