"""Endurance check: 300 hipGraph-replayed steps; asserts steady memory and
finite metrics (catches graph-replay leaks / numeric drift)."""
import sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
from waternet_amd.engine.fast import BenchTrainer

tr = BenchTrainer(batch_size=16, height=112, width=112, device="cuda:0",
                  use_graph=True, seed=5)
for _ in range(20):
    tr.step()
torch.cuda.synchronize()
m0 = torch.cuda.memory_allocated()
import time
t0 = time.perf_counter()
for _ in range(300):
    tr.step()
torch.cuda.synchronize()
dt = time.perf_counter() - t0
m1 = torch.cuda.memory_allocated()
met = tr.metrics()
print(f"img/s={16*300/dt:.1f} mem0={m0/1e6:.1f}MB mem1={m1/1e6:.1f}MB "
      f"peak={torch.cuda.max_memory_allocated()/1e9:.2f}GB metrics={met}")
assert m1 <= m0 + 1_000_000, "memory grew during graph replay"
assert all(abs(v) < 1e12 for v in met.values())
print("ENDURANCE OK")
