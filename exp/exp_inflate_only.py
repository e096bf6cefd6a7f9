"""Inflate-kernel-only run for rocprofv3 (text-like data, Huffman path)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import spark_tfrecord_amd
from spark_tfrecord_amd.io import paths as P
from spark_tfrecord_amd.engine import gpu as g

rng = np.random.default_rng(0)
data = bytes(rng.integers(65, 90, 64 << 20).astype(np.uint8))
os.makedirs("/dev/shm/infonly", exist_ok=True)
p = "/dev/shm/infonly/t.gz"
with open(p, "wb") as f:
    f.write(P.compress_bytes(data, "gzip"))
meta = P.parse_gz_segments_file(p)
total_u = sum(u for _, u in meta[1])
out = torch.empty(total_u, dtype=torch.uint8, device="cuda")
for _ in range(3):
    t0 = time.perf_counter()
    ok = g._device_inflate_group(out, [(p, meta, 0)], torch.device("cuda"))
    torch.cuda.synchronize()
    print(f"inflate {total_u/1e6:.0f}MB ok={ok} {1000*(time.perf_counter()-t0):.1f} ms")
