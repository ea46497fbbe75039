"""BK=64 128-tile vs BK=32 vs hipBLASLt at BERT shapes."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from arkflow_amd.ops import require_native

nat = require_native()


def timeit(fn, reps=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


SHAPES = [  # (M, N, K, act) — BERT-base in-context shapes
    (8192, 3072, 768, 2),   # fc1 + GELU
    (8192, 3072, 768, 0),   # fc1 plain
    (8192, 2304, 768, 0),   # QKV
    (8192, 768, 768, 0),    # attn proj
    (8192, 768, 3072, 0),   # fc2
    (4096, 4096, 4096, 0),  # square reference
]

for M, N, K, act in SHAPES:
    A = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    Bt = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    bias = torch.randn(N, device="cuda")
    bias_h = bias.to(torch.bfloat16)
    ref = A.float() @ Bt.float().T + bias
    if act == 2:
        ref = torch.nn.functional.gelu(ref, approximate="tanh")
    flop = 2 * M * N * K
    rows = [("blaslt", lambda: torch.nn.functional.linear(A, Bt, bias_h))]
    if act != 0:
        rows.append(("blaslt+act", lambda: nat.bias_act_bf16(
            torch.nn.functional.linear(A, Bt, bias_h), None, act)))
    for name, var in [("bk32", 0), ("k64swz", 8), ("k64pre", 10), ("k64wide", 11), ("k64s3", 12),
                      ("8ph", 2), ("8ph128", 5)]:
        rows.append((name,
                     lambda v=var: nat.gemm_bf16_variant(A, Bt, bias, act, v)))
    out = [f"M{M} N{N} K{K} act{act}:"]
    for name, fn in rows:
        try:
            C = fn()
            err = (C.float() - ref).abs().max().item()
            dt = timeit(fn)
            out.append(f"{name} {dt*1e6:.0f}us {flop/dt/1e12:.0f}TF "
                       f"e{err:.2f}")
        except Exception as ex:  # noqa: BLE001
            out.append(f"{name} FAIL({str(ex)[:40]})")
    print("  ".join(out), flush=True)
