"""Phase breakdown of the API read path (flagship schema, 1M rows)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import spark_tfrecord_amd as stf
from bench import make_batch
from spark_tfrecord_amd.engine import gpu as g
from spark_tfrecord_amd.arrow_interop import batch_to_table
from spark_tfrecord_amd.io import paths as P

rows = 1_000_000
batch = make_batch(rows, seed=5)
out = "/dev/shm/apiread2"
os.makedirs(out, exist_ok=True)
g.write_batch_to_file(g.batch_to_device(batch), out + "/part-00000-x.tfrecord",
                      "Example")
files = P.list_data_files(out)

def timed(name, fn, reps=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        r = fn()
    torch.cuda.synchronize()
    print(f"{name:26s} {(time.perf_counter()-t0)/reps*1000:7.2f} ms", flush=True)
    return r

dev_batch, counts = timed("read_files_to_batch", lambda: g.read_files_to_batch(
    files, batch.schema, "Example", verify_crc=True))
host = timed("batch_to_host", lambda: g.batch_to_host(dev_batch))
tbl = timed("batch_to_table", lambda: batch_to_table(host))
timed("full read_tfrecord", lambda: stf.read_tfrecord(out, engine="gpu"),
      reps=5)
timed("read_tfrecord w/ schema", lambda: stf.read_tfrecord(
    out, engine="gpu", schema=batch.schema), reps=5)
print("rows:", tbl.num_rows)
