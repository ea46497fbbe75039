import cProfile, io, os, pstats, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from arkflow_amd.batch import MessageBatch
from arkflow_amd.sql.engine import SqlExecutor

dev = torch.device("cuda:0")
n = 1_000_000
flow = MessageBatch.from_dict({
    **{f"f{i}": torch.rand(n, device=dev) for i in range(16)},
    "key": torch.randint(0, 1024, (n,), dtype=torch.int64, device=dev),
})
ex = SqlExecutor("SELECT key, count(*) AS c, sum(f0) AS s FROM flow "
                 "WHERE f0 >= 0.2 GROUP BY key")
ex.execute({"flow": flow})
torch.cuda.synchronize()
import time
t0 = time.perf_counter()
ex.execute({"flow": flow})
torch.cuda.synchronize()
print("one execute:", time.perf_counter() - t0, flush=True)
pr = cProfile.Profile()
pr.enable()
for _ in range(2):
    ex.execute({"flow": flow})
    torch.cuda.synchronize()
pr.disable()
s = io.StringIO()
pstats.Stats(pr, stream=s).sort_stats("tottime").print_stats(14)
print(s.getvalue()[:2600])
