import os, sys, time
sys.path.insert(0, os.getcwd())
import numpy as np, torch

rng = np.random.default_rng(0)
rows = [rng.standard_normal(784) for _ in range(131072)]

def t(label, fn, n=3):
    fn()
    torch.cuda.synchronize()
    for i in range(n):
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        print(f"{label} run{i}: {(time.perf_counter()-t0)*1000:.1f} ms")

def alloc_pin():
    x = torch.empty((131072, 784), dtype=torch.float64, pin_memory=True)
    return x
t("alloc_pinned_822MB", lambda: alloc_pin())

pinned = alloc_pin()
t("np.stack into pinned", lambda: np.stack(rows, out=pinned.numpy()))
plain = np.empty((131072, 784))
t("np.stack into plain", lambda: np.stack(rows, out=plain))
t("h2d pinned async+sync", lambda: pinned.to("cuda", non_blocking=True))

from sparktorch_amd.utils.data import handle_features_device
from sparktorch_amd.utils.serialize import DataObj
drows = [DataObj(rows[i], 1.0, None, None) for i in range(131072)]
t("handle_features_device", lambda: handle_features_device(drows, 0.0, device="cuda:0"))
