# GPU-side probe: device conversions vs torch e4m3fn, full value sweep
import os, sys, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from brainiak_amd import ops
ext = ops.load_extension()
x = torch.linspace(-500, 500, 2_000_001, device="cuda", dtype=torch.float32)
# add a dense sweep near the z-score range and denormals
x = torch.cat([x, torch.linspace(-2, 2, 4_000_001, device="cuda"),
               torch.linspace(-0.02, 0.02, 400_001, device="cuda")])
ref = x.to(torch.float8_e4m3fn)          # torch (hip) conversion
for sw in (False, True):
    got = ext.debug_fp8_cvt(x, sw=sw)
    gb = got.view(torch.uint8)
    rb = ref.view(torch.uint8)
    neq = (gb != rb)
    # value-level diff (decode both)
    dv = (got.float() - ref.float()).abs()
    n = int(neq.sum())
    print(json.dumps({
        "sw": sw, "n": int(x.numel()), "bytes_differ": n,
        "max_value_diff": float(dv.max()),
        "n_value_diff_gt_ulp": int((dv > 0.13 * x.abs().clamp(min=1)).sum()),
    }), flush=True)
    if n:
        idx = torch.nonzero(neq).flatten()[:10]
        for i in idx.tolist():
            print(f"  x={float(x[i]):.6f} mine=0x{int(gb[i]):02x}"
                  f"({float(got[i].float()):.5f}) torch=0x{int(rb[i]):02x}"
                  f"({float(ref[i].float()):.5f})")
        # worst value diffs
        wi = torch.argsort(dv, descending=True)[:10]
        for i in wi.tolist():
            print(f"  WORST x={float(x[i]):.6f} mine={float(got[i].float()):.5f}"
                  f" torch={float(ref[i].float()):.5f}")
