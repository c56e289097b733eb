"""Schedule-server e2e against the fake k8s API (reference
k8s/src/bin/server.rs endpoints: /apply /delete /listpods /listjobs
/podstatus, each answering with ``execution_results``)."""
import pytest

from tests.test_k8s_operator import fake_api  # noqa: F401  (fixture reuse)


@pytest.fixture()
def client(fake_api):  # noqa: F811
    fastapi = pytest.importorskip("fastapi")  # noqa: F841
    from fastapi.testclient import TestClient

    from persia_amd.k8s_server import create_app

    store, url = fake_api
    with TestClient(create_app(url)) as c:
        yield store, c


def test_apply_list_delete_roundtrip(client):
    store, c = client
    spec = {"gpus_per_node": 8, "data_loader_replicas": 1,
            "metrics_gateway": True}
    r = c.post("/apply", json={
        "job_identifier": {"job_name": "demo", "namespace": "default"},
        "spec": spec,
    }).json()
    assert r["execution_results"]["success"], r
    # trainer Job + 1 loader Job + metrics-gateway Deployment, all labeled
    assert len(store.workloads) == 3
    for m in store.workloads.values():
        assert m["metadata"]["labels"]["persia.ai/job"] == "demo"
    # idempotent re-apply (AlreadyExists swallowed)
    assert c.post("/apply", json={
        "job_identifier": {"job_name": "demo", "namespace": "default"},
        "spec": spec,
    }).json()["execution_results"]["success"]
    assert len(store.workloads) == 3

    r = c.get("/listjobs").json()
    assert r["execution_results"]["success"] and r["resources"] == ["demo"]

    r = c.post("/delete", json={"job_name": "demo"}).json()
    assert r["execution_results"]["success"]
    assert not store.workloads


def test_listpods_and_podstatus(client):
    store, c = client
    store.pods["demo-trainer-abc12"] = {
        "metadata": {"name": "demo-trainer-abc12",
                     "labels": {"persia.ai/job": "demo"}},
        "status": {"phase": "Running"},
    }
    store.pods["other-xyz"] = {
        "metadata": {"name": "other-xyz", "labels": {"persia.ai/job": "other"}},
        "status": {"phase": "Pending"},
    }
    r = c.request("GET", "/listpods", json={"job_name": "demo"}).json()
    assert r["execution_results"]["success"]
    assert r["resources"] == ["demo-trainer-abc12"]

    r = c.request(
        "GET", "/podstatus", json={"pod_name": "demo-trainer-abc12"}
    ).json()
    assert r["execution_results"]["success"] and '"Running"' in r["body"]

    r = c.request("GET", "/podstatus", json={"pod_name": "missing"}).json()
    assert r["execution_results"]["success"] and r["body"] == "None"
