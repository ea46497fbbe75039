"""A/B the 2-phase dbuf 128-tile GEMM at BERT shapes."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from arkflow_amd import ops
nat = ops.require_native()
dev = torch.device("cuda:0")

def bench(fn, iters=50):
    for _ in range(8):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

# refcheck: compare error of v0 (known-good) and v3 side by side
for (M, N, K) in [(512, 512, 64), (8192, 2304, 768), (300, 257, 96)]:
    torch.manual_seed(M + K)
    A = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    Bt = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    ref = A.float() @ Bt.float().T
    e0 = (nat.gemm_bf16_variant(A, Bt, None, 0, 0).float() - ref).abs()
    e3list = [(nat.gemm_bf16_variant(A, Bt, None, 0, 3).float() - ref).abs()
              for _ in range(5)]
    det = all(torch.equal(e3list[0], e) for e in e3list[1:])
    print(f"M{M} N{N} K{K}: v0 max={e0.max():.3f} mean={e0.mean():.5f} | "
          f"v3 max={e3list[0].max():.3f} mean={e3list[0].mean():.5f} "
          f"deterministic={det}", flush=True)
print("2p refcheck compare done", flush=True)

for (M, N, K) in [(8192, 2304, 768), (8192, 3072, 768), (8192, 768, 768),
                  (8192, 768, 3072), (4096, 4096, 4096)]:
    A = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    Bt = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    fl = 2.0 * M * N * K
    t0v = bench(lambda: nat.gemm_bf16_variant(A, Bt, None, 0, 0))
    t3v = bench(lambda: nat.gemm_bf16_variant(A, Bt, None, 0, 3))
    print(f"M{M} N{N} K{K}: 1ph={fl/t0v/1e12:6.1f}TF  2ph={fl/t3v/1e12:6.1f}TF"
          f"  ({t0v/t3v:.2f}x)", flush=True)
