"""Long-horizon stability soak: per-window step timing + allocator water
marks over many thousands of steps (catches drift, leaks, thermal fade)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, torch.nn as nn
from sparktorch_amd.parallel.sync import SyncTrainer

def soak(name, model, x, y, steps, window):
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    tr = SyncTrainer(model, nn.CrossEntropyLoss(), opt, device="cuda:0", world_size=1)
    for _ in range(10):
        tr.train_step(x, y)
    torch.cuda.synchronize()
    torch.cuda.reset_peak_memory_stats()
    base_alloc = torch.cuda.memory_allocated()
    for w in range(steps // window):
        t0 = time.perf_counter()
        for _ in range(window):
            loss = tr.train_step(x, y)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / window * 1000
        print(f"{name} window {w}: {dt:.3f} ms/step loss {loss:.4f} "
              f"alloc {torch.cuda.memory_allocated()/2**30:.2f} GiB "
              f"peak {torch.cuda.max_memory_allocated()/2**30:.2f} GiB", flush=True)
    growth = torch.cuda.memory_allocated() - base_alloc
    print(f"{name}: allocator growth over run = {growth} bytes", flush=True)
    assert abs(growth) < 64 * 2**20, "allocator growth detected"

if __name__ == "__main__":
    torch.manual_seed(0)
    from sparktorch_amd.ops.modules import MnistMLPFused, ResNet18Fused
    x = torch.randn(131072, 784, device="cuda").to(torch.bfloat16)
    y = torch.randint(0, 10, (131072,), device="cuda")
    soak("mlp131k", MnistMLPFused(), x, y, 10000, 2000)
    del x, y; torch.cuda.empty_cache()
    x = torch.randn(256, 3*224*224, device="cuda").to(torch.bfloat16)
    y = torch.randint(0, 1000, (256,), device="cuda")
    soak("resnet256", ResNet18Fused(), x, y, 12000, 2000)
    print("SOAK OK")
