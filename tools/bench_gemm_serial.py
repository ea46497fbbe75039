"""Standalone-vs-serialized GEMM: same kernels, but each iteration's A
depends on the previous C, so consecutive launches cannot overlap. Tests the
hypothesis that pipelined variants only win standalone because back-to-back
independent launches overlap epilogue with the next launch's pipeline fill.
"""
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.join(
    __import__("os").path.dirname(__file__), ".."))
from arkflow_amd import ops

nat = ops.require_native()
dev = torch.device("cuda:0")

M, N, K = 8192, 3072, 768
A = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
Bt = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
fl = 2.0 * M * N * K


def bench(variant, serial, iters=30):
    def one():
        C = nat.gemm_bf16_variant(A, Bt, None, 0, variant)
        if serial:
            # tiny in-place update of A from C: forces launch n+1 to wait
            A[0, 0] += C[0, 0] * 0
    for _ in range(5):
        one()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        one()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return fl / dt / 1e12


for v, name in ((0, "1282"), (3, "2p"), (5, "8p128+swz")):
    free = bench(v, serial=False)
    ser = bench(v, serial=True)
    print(f"{name:10s} independent={free:6.1f} TF   serialized={ser:6.1f} TF"
          f"   ratio={ser/free:.2f}", flush=True)


# cold-weights variant: rotate 12 different Bt tensors (one per BERT layer)
# so weights are L2-cold like the real in-context run
Bts = [torch.randn(N, K, device=dev, dtype=torch.bfloat16) for _ in range(12)]
As = [torch.randn(M, K, device=dev, dtype=torch.bfloat16) for _ in range(4)]


def bench_cold(variant, iters=36):
    def one(i):
        nat.gemm_bf16_variant(As[i % 4], Bts[i % 12], None, 0, variant)
    for i in range(6):
        one(i)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(iters):
        one(i)
    torch.cuda.synchronize()
    return fl / ((time.perf_counter() - t0) / iters) / 1e12


print("-- cold weights (12 rotating Bt, 4 rotating A) --", flush=True)
for v, name in ((0, "1282"), (3, "2p"), (5, "8p128+swz")):
    print(f"{name:10s} cold={bench_cold(v):6.1f} TF", flush=True)


# epilogue variant: bias + GELU (the ACTUAL fc1 configuration in-context)
bias = torch.randn(N, device=dev)


def bench_act(variant, iters=30):
    def one():
        nat.gemm_bf16_variant(A, Bt, bias, 2, variant)
    for _ in range(5):
        one()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        one()
    torch.cuda.synchronize()
    return fl / ((time.perf_counter() - t0) / iters) / 1e12


print("-- bias + GELU epilogue (real fc1 config) --", flush=True)
for v, name in ((0, "1282"), (3, "2p"), (5, "8p128+swz")):
    print(f"{name:10s} gelu={bench_act(v):6.1f} TF", flush=True)
