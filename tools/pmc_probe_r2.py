"""Small kernel probe for PMC capture (rocprofv3 --pmc ... -- this)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch

from arkflow_amd.ops import require_native

nat = require_native()
# k64 GEMM at fc1 shape (fused GELU)
A = torch.randn(8192, 768, device="cuda", dtype=torch.bfloat16)
Bt = torch.randn(3072, 768, device="cuda", dtype=torch.bfloat16)
bias = torch.randn(3072, device="cuda")
for _ in range(10):
    nat.gemm_bf16_variant(A, Bt, bias, 2, 7)
# attention pad=8 at BERT shape
qkv = torch.randn(64, 128, 3, 12, 64, device="cuda", dtype=torch.bfloat16)
for _ in range(10):
    nat.attention_qkv_bf16(qkv, 0.125, 8)
# flash attention S=512
qkv2 = torch.randn(8, 512, 3, 12, 64, device="cuda", dtype=torch.bfloat16)
for _ in range(10):
    nat.attention_qkv_bf16(qkv2, 0.125)
torch.cuda.synchronize()
print("probe done")
