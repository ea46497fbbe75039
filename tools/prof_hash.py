import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from arkflow_amd import ops
nat = ops.require_native()
dev = torch.device("cuda:0")

def t(fn, label, iters=3):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print(f"{label:40} {(time.perf_counter()-t0)/iters*1000:9.2f} ms",
          flush=True)

for n, k in [(8192, 1024), (840_000, 100_000), (840_000, 8192),
             (840_000, 1024), (840_000, 64), (4_000_000, 1024)]:
    keys = torch.randint(0, k, (n,), device=dev, dtype=torch.int64)
    t(lambda: nat.hash_group_i64(keys), f"hash_group n={n} k={k}")
