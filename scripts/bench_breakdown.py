import os, sys, time
sys.path.insert(0, '/root/repo')
from parseable_amd import GpuSession, StandardTableProvider
from parseable_amd.provider import merge_partials
import torch

q = {"select": [{"agg": "count_star"}, {"agg": "max", "col": "latency"}],
     "group_by": ["host"],
     "preds": [{"col": "p_timestamp", "op": "between",
                "lo": 1756684800000, "hi": 1756684800000 + 1908 * 60000}]}
sd = os.environ["GPUQ_DATA"] + "/c2s_1000000000_r0/stream"
sess = GpuSession(device_mask=1)
prov = StandardTableProvider(sd, sess)
plan = prov.scan(dict(q))
plan.load()
for _ in range(3):
    b = plan.execute(0)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    b = plan.execute(0)
torch.cuda.synchronize()
t_exec = (time.perf_counter() - t0) / 10
t0 = time.perf_counter()
for _ in range(10):
    rows = merge_partials([plan.execute(0)], q)
torch.cuda.synchronize()
t_both = (time.perf_counter() - t0) / 10
m = plan.metrics()
print(f"execute-only {t_exec*1e3:.2f} ms; execute+merge {t_both*1e3:.2f} ms; "
      f"kernel_ns/exec {m['kernel_ns']/23/1e6:.2f} exec_ns/exec {m['exec_ns']/23/1e6:.2f}")
plan.close()
