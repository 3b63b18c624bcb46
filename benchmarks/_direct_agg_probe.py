import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

dev = torch.device("cuda:0")
n = 125_000_000
g = torch.Generator(device=dev).manual_seed(1)
keys = torch.randint(0, 1_000_000, (n,), device=dev, generator=g)
vals = torch.rand((n,), device=dev, dtype=torch.float64, generator=g)

def t(name, fn, iters=6):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/iters*1000:.2f} ms", flush=True)

acc = torch.zeros(1_000_000, dtype=torch.float64, device=dev)
t("index_add_ fp64 125M->1M", lambda: acc.index_add_(0, keys, vals))
cnt = torch.zeros(1_000_000, dtype=torch.int64, device=dev)
ones = torch.ones(1, dtype=torch.int64, device=dev).expand(n)
t("bincount 125M->1M", lambda: torch.bincount(keys, minlength=1_000_000))
# smaller range (more contention)
keys2 = torch.randint(0, 10_000, (n,), device=dev, generator=g)
acc2 = torch.zeros(10_000, dtype=torch.float64, device=dev)
t("index_add_ fp64 125M->10k", lambda: acc2.index_add_(0, keys2, vals))
