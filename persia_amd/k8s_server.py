"""HTTP schedule server: apply/delete/inspect Persia jobs over REST.

MI355X-native counterpart of the reference's actix-web scheduler
(/root/reference/k8s/src/bin/server.rs): the same five endpoints with the
same request/response shapes —

* ``POST /apply``     {job_identifier:{job_name,namespace}, spec} — render the
  job's child workloads (persia_amd.k8s_utils.generate_manifests) and create
  them, stamped with the ``persia.ai/job`` ownership label;
* ``POST /delete``    {job_name, namespace} — tear the labeled children down;
* ``GET  /listpods``  {job_name, namespace} — pod names of a job;
* ``GET  /listjobs``  ?namespace= — job names present (PersiaJob CRs and/or
  directly-applied label owners);
* ``GET  /podstatus`` {pod_name, namespace} — raw pod ``status`` JSON.

Every response carries ``execution_results: {success, err_msg}`` exactly like
the reference's ``ExecutionResults`` so existing callers can switch over.

The server is stateless: each request builds a namespace-bound
:class:`persia_amd.k8s_operator.K8sApi` (plain ``requests`` over the k8s REST
API — in-cluster or ``--api-server``), so it is testable against the same
fake API server as the operator (tests/test_k8s_server.py).
"""
from typing import Optional

from pydantic import BaseModel

from persia_amd.k8s_operator import K8sApi
from persia_amd.k8s_utils import generate_manifests
from persia_amd.logger import get_default_logger

_logger = get_default_logger("persia_amd.k8s_server")

OWNER_LABEL = "persia.ai/job"


class JobIdentifier(BaseModel):
    job_name: str
    namespace: str = "default"


class PodIdentifier(BaseModel):
    pod_name: str
    namespace: str = "default"


class ApplyRequest(BaseModel):
    job_identifier: JobIdentifier
    spec: dict = {}


def _ok(**extra) -> dict:
    return {"execution_results": {"success": True, "err_msg": None}, **extra}


def _err(e: Exception, **extra) -> dict:
    return {"execution_results": {"success": False, "err_msg": str(e)}, **extra}


def _labeled_manifests(job_name: str, spec: dict) -> list:
    spec = dict(spec)
    spec["name"] = job_name
    manifests = generate_manifests(spec)
    for m in manifests:
        m["metadata"].setdefault("labels", {})[OWNER_LABEL] = job_name
    return manifests


def create_app(base_url: str, token: Optional[str] = None, verify=True):
    from fastapi import Body, FastAPI

    app = FastAPI(title="persia-schedule-server")

    def api(namespace: str) -> K8sApi:
        return K8sApi(base_url, namespace, token, verify)

    @app.post("/apply")
    def apply(req: ApplyRequest):
        try:
            a = api(req.job_identifier.namespace)
            for m in _labeled_manifests(req.job_identifier.job_name, req.spec):
                a.create_workload(m)  # AlreadyExists is a no-op (idempotent)
            return _ok()
        except Exception as e:
            return _err(e)

    @app.post("/delete")
    def delete(req: JobIdentifier):
        try:
            a = api(req.namespace)
            for kind in ("Job", "Deployment"):
                for wl in a.list_workloads(kind):
                    labels = (wl.get("metadata") or {}).get("labels") or {}
                    if labels.get(OWNER_LABEL) == req.job_name:
                        a.delete_workload(kind, wl["metadata"]["name"])
            return _ok()
        except Exception as e:
            return _err(e)

    @app.get("/listpods")
    def listpods(req: JobIdentifier = Body(...)):
        try:
            pods = api(req.namespace).list_pods(f"{OWNER_LABEL}={req.job_name}")
            return _ok(resources=[p["metadata"]["name"] for p in pods])
        except Exception as e:
            return _err(e, resources=None)

    @app.get("/listjobs")
    def listjobs(namespace: str = "default"):
        try:
            a = api(namespace)
            names = set()
            try:
                names.update(
                    cr["metadata"]["name"] for cr in a.list_persiajobs()
                )
            except Exception:
                pass  # CRD may not be installed when jobs are applied directly
            for kind in ("Job", "Deployment"):
                for wl in a.list_workloads(kind):
                    labels = (wl.get("metadata") or {}).get("labels") or {}
                    if OWNER_LABEL in labels:
                        names.add(labels[OWNER_LABEL])
            return _ok(resources=sorted(names))
        except Exception as e:
            return _err(e, resources=None)

    @app.get("/podstatus")
    def podstatus(req: PodIdentifier = Body(...)):
        try:
            pod = api(req.namespace).get_pod(req.pod_name)
            body = "None" if pod is None else __import__("json").dumps(
                pod.get("status") or {}
            )
            return _ok(body=body)
        except Exception as e:
            return _err(e, body=None)

    return app


def serve(port: int, base_url: str, token: Optional[str] = None, verify=True):
    import uvicorn

    _logger.info(f"schedule server on :{port} -> {base_url}")
    uvicorn.run(create_app(base_url, token, verify), host="0.0.0.0", port=port)
