"""Reconciling-operator e2e against a fake k8s API server (the offline
analog of the reference's k8s system test, k8s/src/bin/e2e.rs:13-60):

* a PersiaJob CR appears -> the operator adds its finalizer, creates the
  child Jobs, and sets status.phase;
* child Jobs succeed -> phase becomes Succeeded;
* the CR is deleted (deletionTimestamp set) -> children are torn down and
  the finalizer released.
"""
import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest

from persia_amd.utils import find_free_port


class FakeK8s:
    """Just enough of the k8s REST API for the operator: CR list/patch with
    a status subresource, and batch/v1 Jobs + apps/v1 Deployments CRUD."""

    def __init__(self):
        self.crs = {}        # name -> PersiaJob dict
        self.workloads = {}  # (kind, name) -> manifest dict
        self.pods = {}       # name -> pod dict (populated by tests)
        self.lock = threading.Lock()

    def handler(self):
        store = self

        class H(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def _send(self, code, obj=None):
                body = json.dumps(obj or {}).encode()
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def _body(self):
                n = int(self.headers.get("Content-Length", 0))
                return json.loads(self.rfile.read(n)) if n else {}

            def do_GET(self):
                from urllib.parse import parse_qs, unquote, urlparse

                u = urlparse(self.path)
                path, qs = u.path, parse_qs(u.query)
                with store.lock:
                    if path.endswith("/persiajobs"):
                        return self._send(200, {"items": list(store.crs.values())})
                    if path.endswith("/pods"):
                        items = list(store.pods.values())
                        sel = unquote(qs.get("labelSelector", [""])[0])
                        if sel:
                            k, v = sel.split("=", 1)
                            items = [
                                p for p in items
                                if (p["metadata"].get("labels") or {}).get(k) == v
                            ]
                        return self._send(200, {"items": items})
                    if "/pods/" in path:
                        pod = store.pods.get(path.rsplit("/", 1)[1])
                        return self._send(200, pod) if pod else self._send(404)
                    for kind, seg in (("Job", "/jobs"), ("Deployment", "/deployments")):
                        if path.endswith(seg):
                            items = [
                                m for (k, _), m in store.workloads.items()
                                if k == kind
                            ]
                            return self._send(200, {"items": items})
                        if seg + "/" in path:
                            name = path.rsplit("/", 1)[1]
                            wl = store.workloads.get((kind, name))
                            return self._send(200, wl) if wl else self._send(404)
                self._send(404)

            def do_POST(self):
                path = self.path.split("?")[0]
                m = self._body()
                with store.lock:
                    if path.endswith("/jobs") or path.endswith("/deployments"):
                        kind = m["kind"]
                        key = (kind, m["metadata"]["name"])
                        if key in store.workloads:
                            return self._send(409)
                        m.setdefault("status", {})
                        store.workloads[key] = m
                        return self._send(201, m)
                self._send(404)

            def do_PATCH(self):
                path = self.path.split("?")[0]
                patch = self._body()
                with store.lock:
                    parts = path.rstrip("/").split("/")
                    sub = None
                    if parts[-1] == "status":
                        sub = "status"
                        parts = parts[:-1]
                    name = parts[-1]
                    if "/persiajobs/" in path:
                        cr = store.crs.get(name)
                        if cr is None:
                            return self._send(404)
                        if sub == "status":
                            cr.setdefault("status", {}).update(patch.get("status", {}))
                        else:
                            for k, v in patch.get("metadata", {}).items():
                                cr["metadata"][k] = v
                        return self._send(200, cr)
                self._send(404)

            def do_DELETE(self):
                path = self.path.split("?")[0]
                with store.lock:
                    for kind, seg in (("Job", "/jobs/"), ("Deployment", "/deployments/")):
                        if seg in path:
                            name = path.rsplit("/", 1)[1]
                            existed = store.workloads.pop((kind, name), None)
                            return self._send(200 if existed else 404)
                self._send(404)

        return H


@pytest.fixture()
def fake_api():
    store = FakeK8s()
    port = find_free_port()
    srv = ThreadingHTTPServer(("127.0.0.1", port), store.handler())
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield store, f"http://127.0.0.1:{port}"
    srv.shutdown()


def test_operator_reconcile_lifecycle(fake_api):
    from persia_amd.k8s_operator import FINALIZER, K8sApi, Operator

    store, url = fake_api
    store.crs["demo"] = {
        "apiVersion": "persia.ai/v1",
        "kind": "PersiaJob",
        "metadata": {"name": "demo"},
        "spec": {"gpus_per_node": 8, "data_loader_replicas": 2,
                 "metrics_gateway": True, "entry": "train.py"},
    }
    op = Operator(K8sApi(url, "default"))

    # 1st reconcile: finalizer added, children created, phase Running
    phases = op.reconcile_all()
    assert FINALIZER in store.crs["demo"]["metadata"]["finalizers"]
    kinds = sorted(k for k, _ in store.workloads)
    assert kinds == ["Deployment", "Job", "Job", "Job"]
    assert ("Job", "demo-trainer") in store.workloads
    assert ("Job", "demo-loader-0") in store.workloads
    assert phases["demo"] == "Running"
    assert store.crs["demo"]["status"]["phase"] == "Running"

    # reconcile is idempotent (no 409 crashes, same children)
    op.reconcile_all()
    assert len(store.workloads) == 4

    # children succeed -> phase Succeeded
    with store.lock:
        for (kind, _name), wl in store.workloads.items():
            if kind == "Job":
                wl["status"] = {"succeeded": 1}
    op.reconcile_all()
    assert store.crs["demo"]["status"]["phase"] == "Succeeded"

    # a failed child -> Failed
    with store.lock:
        store.workloads[("Job", "demo-trainer")]["status"] = {"failed": 1}
    op.reconcile_all()
    assert store.crs["demo"]["status"]["phase"] == "Failed"

    # deletion: children torn down, finalizer released
    with store.lock:
        store.crs["demo"]["metadata"]["deletionTimestamp"] = "2026-09-14T00:00:00Z"
    op.reconcile_all()
    assert store.workloads == {}
    assert store.crs["demo"]["metadata"]["finalizers"] == []


def test_gencrd_roundtrip(capsys):
    import yaml

    from persia_amd.k8s_operator import CRD, main

    main(["gencrd"])
    out = yaml.safe_load(capsys.readouterr().out)
    assert out == CRD
    assert out["spec"]["names"]["kind"] == "PersiaJob"
    assert out["spec"]["versions"][0]["subresources"] == {"status": {}}


def test_operator_broken_cr_does_not_stall_healthy_ones(fake_api):
    """One CR whose spec explodes during manifest rendering must not stop
    the reconcile loop from driving the healthy CRs (k8s_operator
    reconcile_all catches per-CR failures)."""
    from persia_amd.k8s_operator import K8sApi, Operator

    store, url = fake_api
    store.crs["bad"] = {
        "apiVersion": "persia.ai/v1",
        "kind": "PersiaJob",
        "metadata": {"name": "bad"},
        "spec": {"gpus_per_node": "not-a-number"},
    }
    store.crs["good"] = {
        "apiVersion": "persia.ai/v1",
        "kind": "PersiaJob",
        "metadata": {"name": "good"},
        "spec": {"gpus_per_node": 8},
    }
    phases = Operator(K8sApi(url, "default")).reconcile_all()
    assert phases["bad"].startswith("Error:")
    assert phases["good"] == "Running"
    assert ("Job", "good-trainer") in store.workloads
