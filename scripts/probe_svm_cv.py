"""Time ops.svm_cv at the whole-brain shape."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from brainiak_amd.fcma.svm import FoldPlan, svm_cv_device

torch.manual_seed(0)
dev = "cuda:0"
Cvox, E = 34470, 64
base = torch.randn(Cvox, E, 24, device=dev)
K = base @ base.transpose(1, 2) / 24      # PSD kernels
labels = np.tile([0, 1], E // 2)
plan = FoldPlan(labels, 4, dev)
acc = svm_cv_device(K, plan, 1.0, 1e-3)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    svm_cv_device(K, plan, 1.0, 1e-3)
torch.cuda.synchronize()
print("svm_cv %.2f ms  mean acc %.3f" %
      (1e3 * (time.perf_counter() - t0) / 10,
       float(acc.mean())))
