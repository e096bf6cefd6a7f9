"""Phase breakdown of the API write path (flagship schema, 1M rows)."""
import os, sys, time, shutil
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import spark_tfrecord_amd as stf
from bench import make_batch
from spark_tfrecord_amd.engine import gpu as g
from spark_tfrecord_amd.arrow_interop import batch_to_table, table_to_batch
from spark_tfrecord_amd.io import paths as P

rows = 1_000_000
batch = make_batch(rows, seed=6)
table = batch_to_table(batch)
out = "/dev/shm/apiwrite2"

def timed(name, fn, reps=3):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        r = fn()
    torch.cuda.synchronize()
    print(f"{name:28s} {(time.perf_counter()-t0)/reps*1000:8.2f} ms", flush=True)
    return r

wire = timed("table_to_batch", lambda: table_to_batch(table, batch.schema))
dev = timed("batch_to_device", lambda: g.batch_to_device(wire))
img = timed("encode_device", lambda: g.encode_device(dev, "Example"))
view = timed("device_to_pinned_view", lambda: g.device_to_pinned_view(img))
timed("write_file_atomic", lambda: P.write_file_atomic(
    view, "/dev/shm/apiwrite_part.bin"))
timed("FULL write_tfrecord", lambda: stf.write_tfrecord(
    table, out, engine="gpu", mode="overwrite"), reps=3)
# SequenceExample ragged 2-D write (README r1 row)
rng = np.random.default_rng(3)
rag = [[list(rng.random(4).astype(float)) for _ in range(3)]
       for _ in range(200_000)]
seq_tab = {"sid": np.arange(200_000, dtype=np.int64), "rag": rag}
schema = stf.StructType([
    stf.StructField("sid", stf.LongType(), True),
    stf.StructField("rag", stf.ArrayType(stf.ArrayType(stf.FloatType())), True)])
d2 = "/dev/shm/apiwrite_seq"
stf.write_tfrecord(seq_tab, d2, record_type="SequenceExample", schema=schema,
                   engine="gpu", mode="overwrite")
torch.cuda.synchronize()
t0 = time.perf_counter()
stf.write_tfrecord(seq_tab, d2, record_type="SequenceExample", schema=schema,
                   engine="gpu", mode="overwrite")
torch.cuda.synchronize()
print(f"SequenceExample write 200k ragged rows: "
      f"{200_000/(time.perf_counter()-t0)/1e6:.1f}M rows/s")
