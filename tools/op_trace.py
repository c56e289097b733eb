"""Attribute the graphed step's anonymous ATen elementwise kernels: run the
criteo bench with --graph 0 (same kernel sequence, individually launched)
under torch.profiler and print op-level CUDA time grouped by input shape."""
import runpy
import sys

sys.path.insert(0, "/root/repo")
import torch
from torch.profiler import ProfilerActivity, profile

sys.argv = ["bench.py", "--preset", "criteo", "--steps", "10",
            "--warmup", "8", "--graph", "0"]
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             record_shapes=True) as prof:
    try:
        runpy.run_path("/root/repo/bench.py", run_name="__main__")
    except SystemExit:
        pass
print(prof.key_averages(group_by_input_shape=True).table(
    sort_by="self_cuda_time_total", row_limit=45, max_name_column_width=60))
