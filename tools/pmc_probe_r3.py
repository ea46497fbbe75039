"""PMC probe for the round-2-final flagship kernels
(rocprofv3 --pmc ... -- python tools/pmc_probe_r3.py)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch

from arkflow_amd.models.mlp import MlpAnomalyDetector
from arkflow_amd.ops.stepgraph import FusedGenerateFilterInfer

fields = {f"f{i}": {"dtype": "float32", "low": 0.0, "high": 1.0}
          for i in range(16)}
fields["key"] = {"dtype": "int64", "low": 0, "high": 1024}
mlp = MlpAnomalyDetector(16, [256, 256], torch.device("cuda"), 1234)
fused = FusedGenerateFilterInfer(fields, 8192, "f0", ">=", 0.2, mlp,
                                 torch.device("cuda"), seed=7)
for _ in range(30):
    fused.step()
torch.cuda.synchronize()
print("probe done")
