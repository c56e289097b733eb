"""persia-k8s-utils: Kubernetes manifest generation.

The reference ships a Rust operator with a ``PersiaJob`` CRD that deploys the
4-role process topology (k8s/src/crd.rs:42-64).  In the MI355X architecture a
job is one torchrun pod per node (8 ranks over RCCL) plus optional data-loader
pods; this CLI turns a PersiaJob-style yaml spec into plain k8s manifests —
no operator needed.
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


if __name__ == "__main__":
    cli()
