"""LayerNorm microbench: event-timed GB/s at BERT-ish shapes.

Round-2 note: the block-per-row kernel measured 11.9 us avg per dispatch in
the BERT step profile (r2_flagship_kernel_stats.csv) = ~3.2 TB/s effective;
this tool times the dispatched kernel (wave-per-row for n<=2048) directly.
"""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch
from arkflow_amd.ops import require_native


def time_ln(rows, n, iters=200, residual=False):
    nat = require_native()
    x = torch.randn(rows, n, device="cuda").to(torch.bfloat16)
    res = torch.randn(rows, n, device="cuda").to(torch.bfloat16) if residual else None
    g = torch.randn(n, device="cuda")
    b = torch.randn(n, device="cuda")
    for _ in range(20):
        nat.layernorm_bf16(x, g, b, 1e-5, res)
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    torch.cuda.synchronize()
    s.record()
    for _ in range(iters):
        nat.layernorm_bf16(x, g, b, 1e-5, res)
    e.record()
    torch.cuda.synchronize()
    us = s.elapsed_time(e) * 1000 / iters
    nb = rows * n * 2 * (2 if not residual else 3)  # read x (+res), write out
    print(f"rows={rows:7d} n={n:5d} resid={int(residual)}  "
          f"{us:8.2f} us  {nb / us / 1e3:8.1f} GB/s")


if __name__ == "__main__":
    for rows, n in [(4096, 768), (8192, 768), (32768, 768), (8192, 1024),
                    (8192, 4096), (131072, 768)]:
        time_ln(rows, n)
    time_ln(8192, 768, residual=True)
    time_ln(32768, 768, residual=True)
