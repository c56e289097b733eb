"""env/service plumbing (parity with reference test/test_env.py +
test/test_service.py: the two coordinate systems and env-driven service
discovery)."""


def test_nn_worker_env(monkeypatch):
    from persia_amd import env

    monkeypatch.setenv("RANK", "3")
    monkeypatch.setenv("LOCAL_RANK", "1")
    monkeypatch.setenv("WORLD_SIZE", "8")
    assert env.get_rank() == 3
    assert env.get_local_rank() == 1
    assert env.get_world_size() == 8


def test_data_loader_env(monkeypatch):
    from persia_amd import env

    monkeypatch.delenv("RANK", raising=False)
    monkeypatch.setenv("REPLICA_INDEX", "2")
    monkeypatch.setenv("REPLICA_SIZE", "4")
    assert env.get_rank() == 0  # trainer coords default when unset
    assert env.get_replica_index() == 2
    assert env.get_replica_size() == 4


def test_env_defaults(monkeypatch):
    from persia_amd import env

    for name in ("RANK", "LOCAL_RANK", "WORLD_SIZE", "REPLICA_INDEX",
                 "REPLICA_SIZE"):
        monkeypatch.delenv(name, raising=False)
    assert env.get_rank() == 0
    assert env.get_world_size() == 1
    assert env.get_replica_index() == 0
    assert env.get_replica_size() == 1
    assert env.get_master_addr() == "127.0.0.1"


def test_get_embedding_worker_services(monkeypatch):
    from persia_amd.service import get_embedding_worker_services

    monkeypatch.delenv("EMBEDDING_WORKER_SERVICE", raising=False)
    assert get_embedding_worker_services() == []
    monkeypatch.setenv(
        "EMBEDDING_WORKER_SERVICE", "localhost:8887,localhost:8888"
    )
    assert get_embedding_worker_services() == [
        "localhost:8887", "localhost:8888"
    ]
