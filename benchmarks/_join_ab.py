import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from fugue_amd.hip.ext import get_ext
from fugue_amd.hip.ops import _next_pow2

ext = get_ext()
dev = torch.device("cuda:0")
g = torch.Generator(device=dev).manual_seed(0)
nb, np_ = 543_000, 49_000_000
build = torch.randperm(3_000_000, device=dev, generator=g)[:nb].contiguous()
probe = torch.randint(0, 15_000_000, (np_,), device=dev, generator=g)

tsize = _next_pow2(nb * 2)
heads, nxt = ext.join_build(build, tsize)
torch.cuda.synchronize()

def t(name, fn, iters=8):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        r = fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/iters*1000:.2f} ms", flush=True)
    return r

def old_path():
    counts = ext.join_count(probe, build, None, None, heads, nxt, tsize)
    counts64 = counts.to(torch.int64)
    offsets = torch.zeros_like(counts64)
    torch.cumsum(counts64[:-1], 0, out=offsets[1:])
    total = int(counts64.sum().item())
    return ext.join_emit(probe, build, None, None, heads, nxt, tsize, offsets, total, 0)

def new_path():
    return ext.join_pairs(probe, build, None, None, heads, nxt, tsize, 0)

p0, b0 = old_path()
p1, b1 = new_path()
assert p0.numel() == p1.numel(), (p0.numel(), p1.numel())
s0 = torch.argsort(p0 * (1 << 32) + b0); s1 = torch.argsort(p1 * (1 << 32) + b1)
assert torch.equal(p0[s0], p1[s1]) and torch.equal(b0[s0], b1[s1])
print("pairs equal:", p0.numel(), flush=True)
t("old (count+cumsum+emit)", old_path)
t("new (total+chunked emit)", new_path)
t("build", lambda: ext.join_build(build, tsize))
