"""Stability: repeated varied-size API round-trips; RSS + device mem must plateau."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import psutil
import torch
import spark_tfrecord_amd as stf

proc = psutil.Process()
rng = np.random.default_rng(0)
base = "/dev/shm/stress"
os.makedirs(base, exist_ok=True)
t0 = time.perf_counter()
for i in range(60):
    rows = int(rng.integers(1000, 400_000))
    out = f"{base}/d{i % 7}"
    data = {"a": rng.integers(0, 2**50, rows),
            "b": rng.random(rows).astype(np.float32),
            "s": None}
    del data["s"]
    stf.write_tfrecord(data, out, engine="gpu", mode="overwrite",
                       num_shards=1 + i % 3)
    df = stf.read_tfrecord(out, engine="gpu")
    assert df.count() == rows
    if i % 15 == 14:
        rss = proc.memory_info().rss / 1e9
        dev = torch.cuda.memory_allocated() / 1e9
        print(f"iter {i+1}: rss={rss:.2f} GB dev={dev:.3f} GB "
              f"({(time.perf_counter()-t0):.1f}s)")
print("stress ok")
