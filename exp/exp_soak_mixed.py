"""Mixed-workload endurance soak: every major path in one loop for N seconds.

Covers, per iteration: API write (pipelined single shard), API read
(schema-less and explicit), gzip write + device-inflate read, partitionBy
write, schema inference, validate, torch IterableDataset streaming, and an
engine-level round trip. RSS and device memory must stay flat; every read
is count/value-checked. Duration via argv[1] (seconds, default 240).
"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import psutil
import torch
import spark_tfrecord_amd as stf
from spark_tfrecord_amd.torch_data import TFRecordIterableDataset

secs = float(sys.argv[1]) if len(sys.argv) > 1 else 240.0
proc = psutil.Process()
rng = np.random.default_rng(0)
base = "/dev/shm/soakmix"
os.makedirs(base, exist_ok=True)
t0 = time.perf_counter()
it = 0
rss0 = dev0 = None
while time.perf_counter() - t0 < secs:
    it += 1
    rows = int(rng.integers(50_000, 600_000))
    d = f"{base}/d{it % 5}"
    tbl = {
        "id": np.arange(rows, dtype=np.int64),
        "v": rng.random(rows).astype(np.float32),
        "tag": np.array([f"t{i % 13}" for i in range(rows)]),
        "part": np.array([f"p{i % 7}" for i in range(rows)]),
    }
    mode = it % 4
    if mode == 0:  # plain write + schema-less read
        stf.write_tfrecord(tbl, d, engine="gpu", mode="overwrite")
        df = stf.read_tfrecord(d, engine="gpu")
        assert df.count() == rows
    elif mode == 1:  # gzip write + device-inflate read + validate
        stf.write_tfrecord(tbl, d, engine="gpu", mode="overwrite",
                           codec="gzip", num_shards=4)
        df = stf.read_tfrecord(d, engine="gpu")
        assert df.count() == rows
        assert stf.validate_tfrecord(d, engine="gpu").ok
    elif mode == 2:  # partitionBy write + partitioned read + projection
        stf.write_tfrecord(tbl, d, engine="gpu", mode="overwrite",
                           partition_by=["part"])
        df = stf.read_tfrecord(d, engine="gpu", columns=["id", "part"])
        assert df.count() == rows
    else:  # torch streaming + engine round trip
        stf.write_tfrecord(tbl, d, engine="gpu", mode="overwrite")
        ds = TFRecordIterableDataset(d, batch_rows=131072, engine="gpu")
        seen = sum(int(b["_num_rows"]) for b in ds)
        assert seen == rows
    if it % 10 == 0:
        rss = proc.memory_info().rss / 1e9
        dev = torch.cuda.memory_allocated() / 1e9
        if rss0 is None:
            rss0, dev0 = rss, dev
        print(f"iter {it:4d} t={time.perf_counter()-t0:6.1f}s rows={rows} "
              f"rss={rss:.2f}GB dev={dev:.3f}GB", flush=True)
print(f"soak ok: {it} iterations in {time.perf_counter()-t0:.0f}s; "
      f"rss drift {proc.memory_info().rss/1e9 - (rss0 or 0):+.2f} GB",
      flush=True)
