"""Profiling hooks (the reference has none; the rebuild ships
torch.profiler HIP-activity traces plus rocprofv3 recipes).

Usage:
    with trace_step(out_dir="profiles", name="train") as prof:
        trainer.train_step(x, y)
    # writes a chrome trace + a per-op summary table

For per-kernel hardware counters use rocprofv3 on the GPU box (see
tools/bench_conv.py, tools/pmc_conv.py and profiles/*.md for worked runs):
    rocprofv3 --kernel-trace --stats -- python bench.py --steps 5
    rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_INSTS_MFMA,... -- python tools/pmc_conv.py
"""

from __future__ import annotations

import contextlib
import os

import torch


@contextlib.contextmanager
def trace_step(out_dir: str = "profiles", name: str = "step",
               with_stack: bool = False):
    os.makedirs(out_dir, exist_ok=True)
    activities = [torch.profiler.ProfilerActivity.CPU]
    if torch.cuda.is_available():
        activities.append(torch.profiler.ProfilerActivity.CUDA)
    with torch.profiler.profile(activities=activities,
                                with_stack=with_stack) as prof:
        yield prof
    prof.export_chrome_trace(os.path.join(out_dir, f"{name}_trace.json"))
    table = prof.key_averages().table(sort_by="self_cuda_time_total"
                                      if torch.cuda.is_available()
                                      else "self_cpu_time_total",
                                      row_limit=40)
    with open(os.path.join(out_dir, f"{name}_summary.txt"), "w") as f:
        f.write(table)
