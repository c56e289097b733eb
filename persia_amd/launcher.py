"""persia-launcher CLI (mirrors reference persia/launcher.py:106-247).

Roles collapse on an MI355X node: ``nn-worker`` launches the training script
under ``torch.distributed.run`` (one rank per GPU over RCCL); ``data-loader``
launches a loader script with REPLICA_* env.  The reference's
``embedding-worker`` / ``embedding-parameter-server`` binaries have no
standalone equivalent (their state lives inside the trainer ranks) — the
subcommands exist and explain that.
"""
import os
import subprocess
import sys

import click


@click.group()
def cli():
    """persia-launcher"""


@cli.command("nn-worker")
@click.argument("script")
@click.option("--nproc-per-node", default=None, help="ranks per node (default: #GPUs)")
@click.option("--master-addr", default="127.0.0.1")
@click.option("--master-port", default="29500")
@click.argument("extra", nargs=-1)
def nn_worker(script, nproc_per_node, master_addr, master_port, extra):
    script = os.environ.get("PERSIA_NN_WORKER_ENTRY", script)
    if nproc_per_node is None:
        import torch

        nproc_per_node = str(max(1, torch.cuda.device_count()))
    cmd = [
        sys.executable,
        "-m",
        "torch.distributed.run",
        "--nnodes=1",
        f"--nproc-per-node={nproc_per_node}",
        f"--master-addr={master_addr}",
        f"--master-port={master_port}",
        script,
        *extra,
    ]
    sys.exit(subprocess.call(cmd))


@cli.command("data-loader")
@click.argument("script")
@click.option("--replica-index", default="0")
@click.option("--replica-size", default="1")
@click.argument("extra", nargs=-1)
def data_loader(script, replica_index, replica_size, extra):
    script = os.environ.get("PERSIA_DATALOADER_ENTRY", script)
    env = dict(os.environ)
    env["REPLICA_INDEX"] = str(replica_index)
    env["REPLICA_SIZE"] = str(replica_size)
    sys.exit(subprocess.call([sys.executable, script, *extra], env=env))


@cli.command("embedding-worker")
def embedding_worker():
    click.echo(
        "persia_amd has no standalone embedding worker: dedup/lookup/update run "
        "as HIP kernels inside each trainer rank (see persia_amd.core.engine)."
    )


@cli.command("embedding-parameter-server")
def embedding_parameter_server():
    click.echo(
        "persia_amd has no standalone parameter server: the table is sharded "
        "across trainer-rank HBM (see persia_amd.core.store)."
    )


if __name__ == "__main__":
    cli()
