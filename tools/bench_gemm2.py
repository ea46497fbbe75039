"""8-phase GEMM: refcheck (multi-run race screen) + A/B vs 128² and torch."""
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.join(__import__("os").path.dirname(__file__), ".."))
from arkflow_amd import ops

nat = ops.require_native()
dev = torch.device("cuda:0")


def refcheck(M, N, K, variant, act=0, bias=False, runs=3):
    torch.manual_seed(M + N + K + variant)
    A = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    Bt = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    b = torch.randn(N, device=dev) if bias else None
    ref = A.float() @ Bt.float().T
    if bias:
        ref = ref + b
    if act == 2:
        ref = torch.nn.functional.gelu(ref, approximate="tanh")
    outs = []
    for _ in range(runs):
        C = nat.gemm_bf16_variant(A, Bt, b, act, variant)
        outs.append(C)
    torch.cuda.synchronize()
    for i, C in enumerate(outs):
        err = (C.float() - ref).abs().max().item()
        ok = err < 1.0
        if not ok or not torch.equal(outs[0], C):
            print(f"  FAIL v{variant} M{M} N{N} K{K} run{i}: err={err:.3f} "
                  f"deterministic={torch.equal(outs[0], C)}", flush=True)
            return False
    print(f"  ok v{variant} M{M} N{N} K{K} act{act} bias{bias} "
          f"err={err:.4f}", flush=True)
    return True


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


print("=== refcheck ===", flush=True)
ok = True
for variant in (1, 2, 4, 5):
    ok &= refcheck(512, 512, 192, variant)     # NT=3 edge
    ok &= refcheck(8192, 2304, 768, variant)
    ok &= refcheck(8192, 3072, 768, variant, act=2, bias=True)
    ok &= refcheck(8192, 2304, 768, variant, runs=5)  # race screen
    ok &= refcheck(4096, 4096, 4096, variant)
    ok &= refcheck(8200, 2310, 768, variant)   # M,N edges (clamp path)
print("ALL_REFCHECK_OK" if ok else "REFCHECK_FAILED", flush=True)

print("=== perf ===", flush=True)
for (M, N, K) in [(8192, 2304, 768), (8192, 3072, 768), (8192, 768, 768),
                  (8192, 768, 3072), (4096, 4096, 4096)]:
    A = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    Bt = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    B = Bt.T.contiguous()
    fl = 2.0 * M * N * K
    t0v = bench(lambda: nat.gemm_bf16_variant(A, Bt, None, 0, 0))
    t2v = bench(lambda: nat.gemm_bf16_variant(A, Bt, None, 0, 2))
    t5v = bench(lambda: nat.gemm_bf16_variant(A, Bt, None, 0, 5))
    try:
        t3v = bench(lambda: nat.gemm_bf16_variant(A, Bt, None, 0, 3))
    except RuntimeError:
        t3v = float("inf")
    tt = bench(lambda: A @ B)
    print(f"M{M} N{N} K{K}: 128²={fl/t0v/1e12:6.1f}TF  "
          f"8p256+swz={fl/t2v/1e12:6.1f}TF  8p128+swz={fl/t5v/1e12:6.1f}TF  "
          f"2p={fl/t3v/1e12:6.1f}TF  torch={fl/tt/1e12:6.1f}TF", flush=True)
