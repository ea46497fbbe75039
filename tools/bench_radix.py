"""Onesweep radix argsort vs torch(rocPRIM-backed) at 1M/10M keys."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from arkflow_amd.ops import require_native

nat = require_native()


def timeit(fn, reps=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


for n in (1_000_000, 10_000_000):
    for dt in (torch.float32, torch.int64, torch.int32):
        if dt == torch.float32:
            keys = torch.rand(n, device="cuda")
        else:
            keys = torch.randint(0, 1 << 24, (n,), device="cuda", dtype=dt)
        idx = nat.radix_argsort(keys, False)
        ref = torch.argsort(keys, stable=True)
        ok = torch.equal(idx.long(), ref)
        # stability independent check: sorted keys equal & idx is permutation
        perm_ok = torch.equal(torch.sort(idx.long()).values,
                              torch.arange(n, device="cuda"))
        t_ours = timeit(lambda: nat.radix_argsort(keys, False))
        t_torch = timeit(lambda: torch.argsort(keys, stable=True))
        print(f"n={n} {str(dt)[6:]}: ours {t_ours*1e3:.2f} ms  "
              f"torch {t_torch*1e3:.2f} ms  ratio {t_ours/t_torch:.2f}  "
              f"exact={ok} perm={perm_ok}", flush=True)
    # descending spot-check
    keys = torch.rand(1000, device="cuda")
    idx = nat.radix_argsort(keys, True)
    assert torch.equal(keys[idx.long()],
                       torch.sort(keys, descending=True).values)
print("desc ok")
