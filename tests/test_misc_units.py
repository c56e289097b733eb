import numpy as np
import pytest
import torch

from persia_amd.core.schema import EmbeddingSchema, GlobalConfig
from persia_amd.core.storage import PersiaPath
from persia_amd.distributed import BaguaDistributedOption, DDPOption


def test_global_config_from_yaml(tmp_path):
    p = tmp_path / "g.yml"
    p.write_text(
        "common_config:\n  job_type: Train\n  metrics_config:\n"
        "    enable_metrics: true\n  checkpointing_config:\n    num_workers: 4\n"
        "embedding_parameter_server_config:\n  capacity: 12345\n"
    )
    g = GlobalConfig.from_yaml(str(p))
    assert g.job_type == "train"
    assert g.capacity == 12345
    assert g.enable_metrics is True
    assert g.checkpointing_workers == 4


def test_schema_yaml_roundtrip(tmp_path):
    s = EmbeddingSchema.from_dict(
        {
            "feature_index_prefix_bit": 8,
            "slots_config": {
                "a": {"dim": 16, "embedding_summation": False, "sample_fixed_size": 5},
                "b": {"dim": 16, "hash_stack_config": {"hash_stack_rounds": 2,
                                                       "embedding_size": 100}},
            },
        }
    )
    d = s.to_dict()
    s2 = EmbeddingSchema.from_dict(d)
    assert s2.slots["a"].sample_fixed_size == 5
    assert not s2.slots["a"].embedding_summation
    assert s2.slots["b"].hash_stack_config.hash_stack_rounds == 2
    assert s2.slots["a"].index_prefix == s.slots["a"].index_prefix != 0


def test_persia_path_disk(tmp_path):
    p = PersiaPath(str(tmp_path / "sub" / "x.bin"))
    assert not p.is_file()
    p.write_all(b"hello")
    assert p.is_file()
    p.append(b" world")
    assert p.read_to_end() == b"hello world"
    d = PersiaPath(str(tmp_path / "sub"))
    assert any(x.endswith("x.bin") for x in d.list())
    p.remove()
    assert not p.is_file()


def test_bagua_option_mappings():
    for algo in ("gradient_allreduce", "bytegrad", "qadam",
                 "low_precision_decentralized", "decentralized", "async"):
        opt = BaguaDistributedOption(algorithm=algo)
        assert isinstance(opt._ddp, DDPOption)
    assert BaguaDistributedOption(
        algorithm="async", sync_interval_ms=4
    ).sync_every_steps == 4
    with pytest.raises(NotImplementedError):
        BaguaDistributedOption(algorithm="nope")


def test_engine_metrics_noop_when_disabled():
    from persia_amd.core.metrics import EngineMetrics

    m = EngineMetrics(False)
    m.staleness.set(3)
    m.nan_count.inc()
    m.index_miss_count.labels("feat_a").inc(2)


def test_watchdog_dumps_stacks(monkeypatch):
    """PERSIA_DEADLOCK_DETECTION=1 starts the stack-dump thread and a dump
    names live threads (reference parking_lot detector, utils.rs:22-48)."""
    import time

    from persia_amd.core import watchdog

    dumps = []
    monkeypatch.setattr(watchdog, "_started", False)
    monkeypatch.setattr(watchdog._logger, "warning", dumps.append)
    monkeypatch.setenv("PERSIA_DEADLOCK_DETECTION", "1")
    watchdog.maybe_start_deadlock_detection(interval_sec=0.1)
    try:
        for _ in range(100):
            if dumps:
                break
            time.sleep(0.05)
    finally:
        watchdog.stop_deadlock_detection()
    assert dumps, "no stack dump produced"
    assert "watchdog:" in dumps[0] and "MainThread" in dumps[0]
