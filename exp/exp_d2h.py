"""D2H bandwidth vs stream-split count for one large pinned copy."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

N = 200 << 20
dev = torch.empty(N, dtype=torch.uint8, device="cuda")
dev.fill_(7)
host = torch.empty(N, dtype=torch.uint8, device="cpu", pin_memory=True)
main = torch.cuda.current_stream()

for k in (1, 2, 3, 4, 8):
    streams = [torch.cuda.Stream() for _ in range(k)]
    step = (N + k - 1) // k
    def run():
        for s in streams:
            s.wait_stream(main)
        for i, s in enumerate(streams):
            lo, hi = i * step, min(N, (i + 1) * step)
            with torch.cuda.stream(s):
                host[lo:hi].copy_(dev[lo:hi], non_blocking=True)
        for s in streams:
            s.synchronize()
    run()
    t0 = time.perf_counter()
    for _ in range(5):
        run()
    dt = (time.perf_counter() - t0) / 5
    print(f"streams={k}  {dt*1000:7.2f} ms  {N/dt/1e9:6.1f} GB/s", flush=True)

# H2D direction for comparison
for k in (1, 2, 4):
    streams = [torch.cuda.Stream() for _ in range(k)]
    step = (N + k - 1) // k
    def run():
        for s in streams:
            s.wait_stream(main)
        for i, s in enumerate(streams):
            lo, hi = i * step, min(N, (i + 1) * step)
            with torch.cuda.stream(s):
                dev[lo:hi].copy_(host[lo:hi], non_blocking=True)
        for s in streams:
            s.synchronize()
    run()
    t0 = time.perf_counter()
    for _ in range(5):
        run()
    dt = (time.perf_counter() - t0) / 5
    print(f"H2D streams={k}  {dt*1000:7.2f} ms  {N/dt/1e9:6.1f} GB/s", flush=True)

# pinned-allocation cost (caching allocator behavior)
for mb in (1, 8, 64, 200):
    n = mb << 20
    t0 = time.perf_counter()
    for _ in range(5):
        h = torch.empty(n, dtype=torch.uint8, device="cpu", pin_memory=True)
        del h
    print(f"pinned alloc {mb:4d}MB x5: {(time.perf_counter()-t0)*1000:7.2f} ms", flush=True)

# full batch_to_host on a flagship-shaped decode batch
from bench import make_batch
from spark_tfrecord_amd.engine import gpu as g
b = make_batch(1_000_000, seed=6)
dev_b = g.batch_to_device(b)
g.batch_to_host(dev_b)
t0 = time.perf_counter()
for _ in range(5):
    g.batch_to_host(dev_b)
print(f"batch_to_host flagship x5: {(time.perf_counter()-t0)*1000:.1f} ms", flush=True)
