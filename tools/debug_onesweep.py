import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch

from arkflow_amd.ops import require_native

nat = require_native()

# single-effective-pass check: keys < 256 → byte 0 decides everything
for n in (5000, 100_000, 1_000_000):
    keys = torch.randint(0, 256, (n,), device="cuda", dtype=torch.int32)
    idx = nat.radix_argsort(keys, False)
    ref = torch.argsort(keys, stable=True)
    print(f"n={n} byte-keys exact={torch.equal(idx.long(), ref)}")
# tiny full-range
for n in (4096, 8192, 40960):
    keys = torch.rand(n, device="cuda")
    idx = nat.radix_argsort(keys, False)
    ref = torch.argsort(keys, stable=True)
    ok = torch.equal(idx.long(), ref)
    print(f"n={n} f32 exact={ok}")
    if not ok:
        srt = keys[idx.long()]
        bad = (srt[1:] < srt[:-1]).nonzero()
        print("  first disorder at", bad[:5].flatten().tolist(),
              "permcheck", torch.equal(torch.sort(idx.long()).values,
                                       torch.arange(n, device="cuda")))
