"""Per-stage wall-clock tracing (SURVEY §5 tracing parity).

The reference logs ``time.time()`` deltas around each pipeline stage
(ref src/brainiak/fcma/voxelselector.py:299-327, classifier.py:459-503);
here the same observability comes from one context manager, with two
MI355X extras:

- GPU stages bracket with ``torch.cuda.synchronize()`` so the logged
  time is real device time, not launch time (opt-in via ``sync=``);
- ``BRAINIAK_TORCH_PROFILE=<dir>`` wraps the region in
  ``torch.profiler`` and drops a chrome trace under <dir> — the local
  complement to the rocprofv3 recipes in profiles/README.md.
"""

import logging
import os
import time
from contextlib import contextmanager

__all__ = ["stage_timer"]


@contextmanager
def stage_timer(stage, logger=None, sync_device=None):
    """Log the wall-clock of a pipeline stage at INFO.

    sync_device: pass a torch device (or device string) to synchronize
    before both timestamps — required for honest GPU stage timings.
    """
    log = logger or logging.getLogger(__name__)
    profile_dir = os.environ.get("BRAINIAK_TORCH_PROFILE")

    def _sync():
        if sync_device is None:
            return
        import torch
        if torch.device(sync_device).type == "cuda" \
                and torch.cuda.is_available():
            torch.cuda.synchronize(sync_device)

    prof = None
    if profile_dir:
        import torch.profiler as tp
        acts = [tp.ProfilerActivity.CPU]
        import torch
        if torch.cuda.is_available():
            acts.append(tp.ProfilerActivity.CUDA)
        prof = tp.profile(activities=acts)
        prof.__enter__()
    _sync()
    begin = time.perf_counter()
    try:
        yield
    finally:
        _sync()
        elapsed = time.perf_counter() - begin
        log.info("%s took %.3f s", stage, elapsed)
        if prof is not None:
            prof.__exit__(None, None, None)
            os.makedirs(profile_dir, exist_ok=True)
            out = os.path.join(
                profile_dir,
                "%s_%d.json" % (stage.replace(" ", "_"),
                                int(time.time() * 1e3)))
            prof.export_chrome_trace(out)
            log.info("torch.profiler trace for '%s' -> %s", stage, out)
