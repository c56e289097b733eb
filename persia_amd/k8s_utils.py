"""persia-k8s-utils: Kubernetes manifest generation.

The reference ships a Rust operator with a ``PersiaJob`` CRD that deploys the
4-role process topology (k8s/src/crd.rs:42-64).  In the MI355X architecture a
job is one torchrun pod per node (8 ranks over RCCL) plus optional data-loader
pods.  ``gen`` renders the manifests for manual apply; ``gencrd`` and
``operator`` (persia_amd/k8s_operator.py) provide the reconciling-controller
workflow: create PersiaJob CRs and the operator creates/patches these same
workloads and reflects status.
"""
import sys

import click
import yaml


def generate_manifests(spec: dict) -> list:
    name = spec.get("name", "persia-job")
    image = spec.get("image", "rocm/pytorch:latest")
    env = [{"name": k, "value": str(v)} for k, v in spec.get("env", {}).items()]
    gpus = int(spec.get("gpus_per_node", 8))
    trainer = {
        "apiVersion": "batch/v1",
        "kind": "Job",
        "metadata": {"name": f"{name}-trainer"},
        "spec": {
            "backoffLimit": 0,
            "template": {
                "spec": {
                    "restartPolicy": "Never",
                    "containers": [
                        {
                            "name": "trainer",
                            "image": image,
                            "command": [
                                "python", "-m", "torch.distributed.run",
                                "--nnodes=1", f"--nproc-per-node={gpus}",
                                "--master-addr=127.0.0.1",
                                spec.get("entry", "train.py"),
                            ],
                            "env": env
                            + [{"name": "HSA_ENABLE_IPC_MODE_LEGACY", "value": "0"}],
                            "resources": {"limits": {"amd.com/gpu": gpus}},
                        }
                    ],
                }
            },
        },
    }
    out = [trainer]
    n_loaders = int(spec.get("data_loader_replicas", 0))
    for i in range(n_loaders):
        out.append(
            {
                "apiVersion": "batch/v1",
                "kind": "Job",
                "metadata": {"name": f"{name}-loader-{i}"},
                "spec": {
                    "template": {
                        "spec": {
                            "restartPolicy": "Never",
                            "containers": [
                                {
                                    "name": "loader",
                                    "image": image,
                                    "command": [
                                        "python",
                                        spec.get("data_loader_entry", "data_loader.py"),
                                    ],
                                    "env": env
                                    + [
                                        {"name": "REPLICA_INDEX", "value": str(i)},
                                        {"name": "REPLICA_SIZE", "value": str(n_loaders)},
                                    ],
                                }
                            ],
                        }
                    }
                },
            }
        )
    if spec.get("metrics_gateway", False):
        out.append(
            {
                "apiVersion": "apps/v1",
                "kind": "Deployment",
                "metadata": {"name": f"{name}-metrics-gateway"},
                "spec": {
                    "replicas": 1,
                    "selector": {"matchLabels": {"app": f"{name}-gw"}},
                    "template": {
                        "metadata": {"labels": {"app": f"{name}-gw"}},
                        "spec": {
                            "containers": [
                                {"name": "gw", "image": "prom/pushgateway",
                                 "ports": [{"containerPort": 9091}]}
                            ]
                        },
                    },
                },
            }
        )
    return out


@click.group()
def cli():
    """persia-k8s-utils"""


@cli.command("gen")
@click.argument("spec_file")
def gen(spec_file):
    """Generate k8s manifests from a PersiaJob-style yaml spec."""
    with open(spec_file, "r", encoding="utf-8") as f:
        spec = yaml.safe_load(f)
    print(yaml.safe_dump_all(generate_manifests(spec)))


@cli.command("gencrd")
def gencrd():
    """Print the PersiaJob CRD yaml (reference k8s/src/bin/gencrd.rs)."""
    from persia_amd.k8s_operator import main as op_main

    op_main(["gencrd"])


@cli.command("operator")
@click.option("--api-server", default=None, help="k8s API base URL (default: in-cluster)")
@click.option("--namespace", default=None)
@click.option("--token", default=None)
@click.option("--period-sec", default=2.0, type=float)
def operator(api_server, namespace, token, period_sec):
    """Run the PersiaJob reconciling operator (k8s/src/bin/operator.rs)."""
    from persia_amd.k8s_operator import main as op_main

    argv = ["operator", f"--period-sec={period_sec}"]
    if api_server:
        argv.append(f"--api-server={api_server}")
    if namespace:
        argv.append(f"--namespace={namespace}")
    if token:
        argv.append(f"--token={token}")
    op_main(argv)


@cli.command("server")
@click.option("--port", required=True, type=int)
@click.option("--api-server", default=None, help="k8s API base URL (default: in-cluster)")
@click.option("--token", default=None)
def server(port, api_server, token):
    """Run the HTTP schedule server (reference k8s/src/bin/server.rs)."""
    from persia_amd.k8s_operator import main as op_main

    argv = ["server", f"--port={port}"]
    if api_server:
        argv.append(f"--api-server={api_server}")
    if token:
        argv.append(f"--token={token}")
    op_main(argv)


if __name__ == "__main__":
    cli()
