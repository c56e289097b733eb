"""PersiaJob operator: a reconciling controller for Kubernetes.

MI355X-native counterpart of the reference's Rust operator
(/root/reference/k8s/src/crd.rs:42-64, k8s/src/bin/operator.rs,
k8s/src/finalizer.rs): it watches ``PersiaJob`` custom resources and drives
the child workloads that ``persia_amd.k8s_utils.generate_manifests`` renders
(one torchrun trainer Job per node + optional loader Jobs + metrics gateway),
reflecting child state back into ``status.phase`` and cleaning up through a
finalizer on deletion.

Design differences from the reference, on purpose:

* **level-triggered reconcile loop** over the plain k8s REST API (polling
  ``resourceVersion``-free lists) instead of a watch stream — the observed
  state is re-derived every period, so a missed event can never wedge a job;
* **no client library**: the handful of endpoints used (list/patch the CR,
  get/create/delete Jobs) speak JSON over ``requests``, which keeps the
  operator runnable in this offline image and trivially testable against a
  fake API server (tests/test_k8s_operator.py).

Runs in-cluster (service-account token + CA) or against an explicit
``--api-server`` (tests, kubectl proxy).
"""
import json
import os
import time
from typing import Dict, List, Optional

from persia_amd.k8s_utils import generate_manifests
from persia_amd.logger import get_default_logger

_logger = get_default_logger("persia_amd.k8s_operator")

GROUP = "persia.ai"
VERSION = "v1"
PLURAL = "persiajobs"
FINALIZER = "persia.ai/cleanup"

CRD = {
    "apiVersion": "apiextensions.k8s.io/v1",
    "kind": "CustomResourceDefinition",
    "metadata": {"name": f"{PLURAL}.{GROUP}"},
    "spec": {
        "group": GROUP,
        "names": {
            "kind": "PersiaJob",
            "plural": PLURAL,
            "singular": "persiajob",
            "shortNames": ["pj"],
        },
        "scope": "Namespaced",
        "versions": [
            {
                "name": VERSION,
                "served": True,
                "storage": True,
                "subresources": {"status": {}},
                "schema": {
                    "openAPIV3Schema": {
                        "type": "object",
                        "properties": {
                            "spec": {
                                "type": "object",
                                "x-kubernetes-preserve-unknown-fields": True,
                            },
                            "status": {
                                "type": "object",
                                "x-kubernetes-preserve-unknown-fields": True,
                            },
                        },
                    }
                },
            }
        ],
    },
}


class K8sApi:
    """Minimal typed wrapper over the k8s REST endpoints the operator uses."""

    def __init__(self, base_url: str, namespace: str = "default",
                 token: Optional[str] = None, verify=True):
        import requests

        self.base = base_url.rstrip("/")
        self.ns = namespace
        self.s = requests.Session()
        self.s.verify = verify
        if token:
            self.s.headers["Authorization"] = f"Bearer {token}"

    @staticmethod
    def in_cluster(namespace: Optional[str] = None) -> "K8sApi":
        sa = "/var/run/secrets/kubernetes.io/serviceaccount"
        with open(f"{sa}/token", encoding="utf-8") as f:
            token = f.read()
        ns = namespace
        if ns is None:
            with open(f"{sa}/namespace", encoding="utf-8") as f:
                ns = f.read().strip()
        host = os.environ["KUBERNETES_SERVICE_HOST"]
        port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        return K8sApi(f"https://{host}:{port}", ns, token, verify=f"{sa}/ca.crt")

    # ---- PersiaJob CRs ----
    def _cr_base(self) -> str:
        return f"{self.base}/apis/{GROUP}/{VERSION}/namespaces/{self.ns}/{PLURAL}"

    def list_persiajobs(self) -> List[dict]:
        r = self.s.get(self._cr_base())
        r.raise_for_status()
        return r.json().get("items", [])

    def patch_persiajob(self, name: str, patch: dict, subresource: str = "") -> None:
        url = f"{self._cr_base()}/{name}{'/' + subresource if subresource else ''}"
        r = self.s.patch(
            url, data=json.dumps(patch),
            headers={"Content-Type": "application/merge-patch+json"},
        )
        r.raise_for_status()

    # ---- child workloads (batch Jobs + apps Deployments) ----
    def _wl_base(self, kind: str) -> str:
        if kind == "Job":
            return f"{self.base}/apis/batch/v1/namespaces/{self.ns}/jobs"
        if kind == "Deployment":
            return f"{self.base}/apis/apps/v1/namespaces/{self.ns}/deployments"
        raise ValueError(kind)

    def get_workload(self, kind: str, name: str) -> Optional[dict]:
        r = self.s.get(f"{self._wl_base(kind)}/{name}")
        if r.status_code == 404:
            return None
        r.raise_for_status()
        return r.json()

    def create_workload(self, manifest: dict) -> None:
        r = self.s.post(
            self._wl_base(manifest["kind"]), data=json.dumps(manifest),
            headers={"Content-Type": "application/json"},
        )
        if r.status_code != 409:  # AlreadyExists is a reconcile no-op
            r.raise_for_status()

    def delete_workload(self, kind: str, name: str) -> None:
        r = self.s.delete(
            f"{self._wl_base(kind)}/{name}",
            params={"propagationPolicy": "Background"},
        )
        if r.status_code != 404:
            r.raise_for_status()

    def list_workloads(self, kind: str) -> List[dict]:
        r = self.s.get(self._wl_base(kind))
        r.raise_for_status()
        return r.json().get("items", [])

    # ---- pods (read-only; used by the schedule server) ----
    def _pod_base(self) -> str:
        return f"{self.base}/api/v1/namespaces/{self.ns}/pods"

    def list_pods(self, label_selector: Optional[str] = None) -> List[dict]:
        params = {"labelSelector": label_selector} if label_selector else None
        r = self.s.get(self._pod_base(), params=params)
        r.raise_for_status()
        return r.json().get("items", [])

    def get_pod(self, name: str) -> Optional[dict]:
        r = self.s.get(f"{self._pod_base()}/{name}")
        if r.status_code == 404:
            return None
        r.raise_for_status()
        return r.json()


def _owned_manifests(job: dict) -> List[dict]:
    """Render the child workloads for a PersiaJob CR and stamp ownership
    labels (the reference stamps ownerReferences; labels keep the fake-server
    tests simple and GC is done by the finalizer path anyway)."""
    spec = dict(job.get("spec") or {})
    spec.setdefault("name", job["metadata"]["name"])
    manifests = generate_manifests(spec)
    for m in manifests:
        labels = m["metadata"].setdefault("labels", {})
        labels["persia.ai/job"] = job["metadata"]["name"]
    return manifests


def _job_phase(children: List[Optional[dict]]) -> str:
    """Aggregate child Job statuses into a PersiaJob phase (the reference
    watches nn-worker pod completion the same way, k8s/src/bin/e2e.rs)."""
    if any(c is None for c in children):
        return "Pending"
    statuses = [c.get("status") or {} for c in children]
    if any(s.get("failed") for s in statuses):
        return "Failed"
    jobs = [c for c in children if c.get("kind") == "Job"]
    if jobs and all((c.get("status") or {}).get("succeeded") for c in jobs):
        return "Succeeded"
    return "Running"


class Operator:
    def __init__(self, api: K8sApi, period_sec: float = 2.0):
        self.api = api
        self.period = period_sec
        self._stop = False

    def reconcile_one(self, job: dict) -> str:
        name = job["metadata"]["name"]
        finalizers = job["metadata"].get("finalizers") or []
        if job["metadata"].get("deletionTimestamp"):
            # deletion: tear down children, then release our finalizer
            for m in _owned_manifests(job):
                self.api.delete_workload(m["kind"], m["metadata"]["name"])
            if FINALIZER in finalizers:
                finalizers = [f for f in finalizers if f != FINALIZER]
                self.api.patch_persiajob(
                    name, {"metadata": {"finalizers": finalizers}}
                )
            return "Deleted"
        if FINALIZER not in finalizers:
            self.api.patch_persiajob(
                name, {"metadata": {"finalizers": finalizers + [FINALIZER]}}
            )
        children = []
        for m in _owned_manifests(job):
            existing = self.api.get_workload(m["kind"], m["metadata"]["name"])
            if existing is None:
                _logger.info(f"{name}: creating {m['kind']} {m['metadata']['name']}")
                self.api.create_workload(m)
                existing = self.api.get_workload(m["kind"], m["metadata"]["name"])
            children.append(existing)
        phase = _job_phase(children)
        if (job.get("status") or {}).get("phase") != phase:
            _logger.info(f"{name}: phase -> {phase}")
            self.api.patch_persiajob(
                name, {"status": {"phase": phase}}, subresource="status"
            )
        return phase

    def reconcile_all(self) -> Dict[str, str]:
        out = {}
        for job in self.api.list_persiajobs():
            name = job["metadata"]["name"]
            try:
                out[name] = self.reconcile_one(job)
            except Exception as e:  # one broken CR must not stall the rest
                _logger.warning(f"reconcile {name} failed: {e}")
                out[name] = f"Error: {e}"
        return out

    def run_forever(self):
        _logger.info(f"operator watching {self.api.ns}/{PLURAL} every {self.period}s")
        while not self._stop:
            self.reconcile_all()
            time.sleep(self.period)

    def stop(self):
        self._stop = True


def main(argv: Optional[List[str]] = None):
    import argparse

    p = argparse.ArgumentParser(prog="persia-k8s-operator")
    sub = p.add_subparsers(dest="cmd", required=True)
    sub.add_parser("gencrd", help="print the PersiaJob CRD yaml")
    op = sub.add_parser("operator", help="run the reconcile loop")
    op.add_argument("--api-server", default=None,
                    help="k8s API base URL (default: in-cluster config)")
    op.add_argument("--namespace", default=None)
    op.add_argument("--token", default=None)
    op.add_argument("--period-sec", type=float, default=2.0)
    sv = sub.add_parser(
        "server", help="HTTP schedule server (apply/delete/list/podstatus)"
    )
    sv.add_argument("--port", type=int, required=True)
    sv.add_argument("--api-server", default=None,
                    help="k8s API base URL (default: in-cluster config)")
    sv.add_argument("--token", default=None)
    args = p.parse_args(argv)
    if args.cmd == "gencrd":
        import yaml

        print(yaml.safe_dump(CRD))
        return
    if args.cmd == "server":
        from persia_amd.k8s_server import serve

        if args.api_server:
            serve(args.port, args.api_server, args.token)
        else:
            sa = "/var/run/secrets/kubernetes.io/serviceaccount"
            with open(f"{sa}/token", encoding="utf-8") as f:
                token = f.read()
            host = os.environ["KUBERNETES_SERVICE_HOST"]
            kport = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            serve(args.port, f"https://{host}:{kport}", token,
                  verify=f"{sa}/ca.crt")
        return
    if args.api_server:
        api = K8sApi(args.api_server, args.namespace or "default", args.token)
    else:
        api = K8sApi.in_cluster(args.namespace)
    Operator(api, period_sec=args.period_sec).run_forever()


if __name__ == "__main__":
    main()
