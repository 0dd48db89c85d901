import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, torch
from sparktorch_amd import ops
ext = ops.ext()
def t(fn, iters=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6
shapes = [  # B, CI, CO, H (resnet l1..l4 @ bench B=256)
    (256, 64, 64, 56), (256, 128, 128, 28), (256, 256, 256, 14), (256, 512, 512, 7),
]
for B, CI, CO, H in shapes:
    x = torch.randn(B, H, W := H, CI, device="cuda").to(torch.bfloat16).contiguous()
    dz = torch.randn(B * H * W, CO, device="cuda").to(torch.bfloat16).contiguous()
    xP = ext.pad_nhwc(x, 1)
    fl = 2.0 * CO * 9 * CI * B * H * W
    print(f"CI={CI} CO={CO} R={B*H*W}")
    for sk in (16, 64, 128, 256, 409, 512):
        if sk * 64 > B * H * W: continue
        ta = t(lambda: ext.conv_implicit_wgrad(dz, xP, 3, 3, sk, False))
        tb = t(lambda: ext.conv_implicit_wgrad(dz, xP, 3, 3, sk, True))
        print(f"  sk{sk:<4} atomic {ta:8.1f}us ({fl/ta/1e6:4.0f} TF)  slab {tb:8.1f}us ({fl/tb/1e6:4.0f} TF)")
