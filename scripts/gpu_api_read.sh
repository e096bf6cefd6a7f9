cd "$GRAFT_REPO_ROOT"
timeout 420 python -m pytest tests -m gpu -x -q 2>&1 | tail -1
timeout 300 python - <<'PY' 2>&1 | tail -3
import time
import numpy as np
import spark_tfrecord_amd as stf
from bench import make_batch
from spark_tfrecord_amd.engine import gpu as g
from spark_tfrecord_amd.arrow_interop import batch_to_table

rows = 1_000_000
batch = make_batch(rows, seed=5)
out = "/dev/shm/apiread/t"
import os; os.makedirs(out, exist_ok=True)
g.write_batch_to_file(g.batch_to_device(batch), out + "/part-00000-x.tfrecord", "Example")
stf.read_tfrecord(out, engine="gpu")  # warm
t0 = time.perf_counter()
df = stf.read_tfrecord(out, engine="gpu")
t1 = time.perf_counter()
assert df.count() == rows
print(f"API read (flagship schema, arrow out): {rows/(t1-t0)/1e6:.1f}M rows/s")
# device-resident read (torch dataset path)
from spark_tfrecord_amd.torch_data import TFRecordIterableDataset
ds = TFRecordIterableDataset(out, batch_rows=10**9, engine="gpu")
import torch
next(iter(ds)); torch.cuda.synchronize()
t0 = time.perf_counter()
total = sum(int(b["_num_rows"]) for b in ds)
torch.cuda.synchronize()
t1 = time.perf_counter()
print(f"device-resident stream: {total/(t1-t0)/1e6:.1f}M rows/s")
PY
timeout 300 python - <<'PY' 2>&1 | tail -2
import time, os
import numpy as np
import spark_tfrecord_amd as stf
from bench import make_batch
from spark_tfrecord_amd.arrow_interop import batch_to_table

rows = 1_000_000
table = batch_to_table(make_batch(rows, seed=6))
out = "/dev/shm/apiwrite/t"
stf.write_tfrecord(table, out, engine="gpu", mode="overwrite")  # warm
t0 = time.perf_counter()
stf.write_tfrecord(table, out, engine="gpu", mode="overwrite")
t1 = time.perf_counter()
print(f"API write (flagship schema, arrow in): {rows/(t1-t0)/1e6:.1f}M rows/s")
PY
