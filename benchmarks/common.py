"""Shared benchmark plumbing: distributed init, timing, JSON emission."""

import json
import os
import time

import numpy as np
import torch


def dist_setup():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if torch.cuda.is_available():
        local = int(os.environ.get("LOCAL_RANK", rank))
        device = torch.device("cuda", local % torch.cuda.device_count())
        torch.cuda.set_device(device)
        backend = "nccl"
    else:
        device = torch.device("cpu")
        backend = "gloo"
    if world > 1:
        import torch.distributed as dist
        dist.init_process_group(backend=backend)
    return rank, world, device, backend


def barrier_sync(world, device):
    if world > 1:
        import torch.distributed as dist
        dist.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize(device)


def timed_steps(fn, steps, warmup, world, device):
    for i in range(warmup):
        fn(i)
    barrier_sync(world, device)
    t0 = time.perf_counter()
    for i in range(steps):
        fn(warmup + i)
    barrier_sync(world, device)
    elapsed = time.perf_counter() - t0
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if device.type == "cuda" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])
    return elapsed


def emit(rank, metric, value, unit, world, steps, warmup, elapsed,
         higher_is_better, scaling, dtype, config):
    if rank == 0:
        print(json.dumps({
            "metric": metric, "value": value, "unit": unit,
            "n_gpus": world, "steps": steps, "warmup": warmup,
            "ms_per_step": elapsed * 1000.0 / steps,
            "higher_is_better": higher_is_better, "scaling": scaling,
            "vs_baseline": None, "dtype": dtype, "data": "synthetic",
            "config": config}))


def teardown(world):
    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()
