import yaml

from persia_amd.k8s_utils import generate_manifests


def test_generate_manifests():
    spec = {
        "name": "dlrm-train",
        "image": "rocm/persia:latest",
        "gpus_per_node": 8,
        "entry": "train.py",
        "data_loader_replicas": 2,
        "metrics_gateway": True,
        "env": {"LOG_LEVEL": "INFO"},
    }
    out = generate_manifests(spec)
    assert len(out) == 4  # trainer + 2 loaders + gateway
    trainer = out[0]
    cmd = trainer["spec"]["template"]["spec"]["containers"][0]["command"]
    assert "--nproc-per-node=8" in cmd
    assert trainer["spec"]["template"]["spec"]["containers"][0]["resources"]["limits"][
        "amd.com/gpu"
    ] == 8
    loader1 = out[2]
    env = {e["name"]: e["value"] for e in loader1["spec"]["template"]["spec"]["containers"][0]["env"]}
    assert env["REPLICA_INDEX"] == "1"
    assert env["REPLICA_SIZE"] == "2"
    # round-trips through yaml
    yaml.safe_dump_all(out)


def test_launcher_cli_help():
    from click.testing import CliRunner

    from persia_amd.launcher import cli

    r = CliRunner().invoke(cli, ["--help"])
    assert r.exit_code == 0
    for sub in ("nn-worker", "data-loader", "embedding-worker"):
        assert sub in r.output
