"""RCCL-on-gfx950 evidence probe (single rank — the box has one GPU).

Initializes torch.distributed with backend nccl (= RCCL on ROCm), runs
all_reduce / all_gather / all_to_all_single / reduce_scatter on device
tensors, and exercises the repartition collective path end to end. One
rank per GPU is exactly the production topology; this proves the RCCL
library loads, builds its communicator on gfx950, and the collective
call-sites are correct. Multi-rank behavior is CPU-CI-tested over gloo
(world 2-4) with identical code.
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import torch.distributed as dist

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29531")
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")
dist.init_process_group("nccl")
torch.cuda.set_device(0)
dev = torch.device("cuda:0")

x = torch.randn(1 << 20, device=dev)
ref = x.clone()
dist.all_reduce(x)
assert torch.equal(x, ref), "world-1 all_reduce must be identity"
out = [torch.empty_like(x)]
dist.all_gather(out, x)
assert torch.equal(out[0], x)
y = torch.empty_like(x)
dist.all_to_all_single(y, x)
assert torch.equal(y, x)
rs = torch.empty_like(x)
dist.reduce_scatter_tensor(rs, x)
assert torch.equal(rs, x)

from arkflow_amd.batch import Column, MessageBatch
from arkflow_amd.parallel import dist as afdist

batch = MessageBatch({
    "k": Column("numeric", torch.randint(0, 64, (8192,), device=dev,
                                         dtype=torch.int64)),
    "v": Column("numeric", torch.rand(8192, device=dev)),
})
outb = afdist.repartition_by_key(batch, "k")
assert outb.num_rows == 8192 and outb.column("v").data.is_cuda
print("RCCL probe ok: all_reduce/all_gather/all_to_all/reduce_scatter +"
      " repartition on gfx950, backend", dist.get_backend())
dist.destroy_process_group()
