"""cProfile of the flagship bench step loop (host-side orchestration)."""
import cProfile, os, pstats, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from bench import make_batch
from spark_tfrecord_amd.engine import gpu as g

batch = make_batch(1_000_000, seed=1)
dev = g.batch_to_device(batch)
path = "/dev/shm/stepprof/bench.tfrecord"
os.makedirs(os.path.dirname(path), exist_ok=True)

def step():
    g.write_batch_to_file(dev, path, "Example")
    return g.read_file_to_batch_pipelined(path, batch.schema, "Example", True)

for _ in range(5):
    step()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    step()
torch.cuda.synchronize()
print(f"{(time.perf_counter()-t0)/10*1000:.2f} ms/step")
pr = cProfile.Profile()
pr.enable()
for _ in range(10):
    step()
torch.cuda.synchronize()
pr.disable()
pstats.Stats(pr).sort_stats("tottime").print_stats(24)
