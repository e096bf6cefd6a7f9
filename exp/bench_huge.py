"""Profile target: the 16KB-record shape only (roadmap item: large records
starve the chip). Phase-split timings + enough steps for rocprofv3 stats."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import spark_tfrecord_amd as stf
from spark_tfrecord_amd.columnar import RecordBatch, WireColumn
from spark_tfrecord_amd.schema import KIND_BYTES, KIND_FLOAT, KIND_INT64
from spark_tfrecord_amd.engine import gpu as g

rows, floats, sb = 16_000, 256, 15_000
rng = np.random.default_rng(0)
fields = [stf.StructField("id", stf.LongType(), True),
          stf.StructField("floats", stf.ArrayType(stf.FloatType()), True),
          stf.StructField("blob", stf.StringType(), True)]
cols = [WireColumn(KIND_INT64, False, np.ones(rows, np.uint8),
                   np.arange(rows + 1, dtype=np.int64),
                   rng.integers(0, 2**62, rows).astype(np.int64)),
        WireColumn(KIND_FLOAT, False, np.ones(rows, np.uint8),
                   np.arange(0, (rows+1)*floats, floats, dtype=np.int64),
                   rng.random(rows*floats).astype(np.float32)),
        WireColumn(KIND_BYTES, False, np.ones(rows, np.uint8),
                   np.arange(rows + 1, dtype=np.int64),
                   rng.integers(65, 90, rows*sb).astype(np.uint8),
                   elem_off=np.arange(0, (rows+1)*sb, sb, dtype=np.int64))]
batch = RecordBatch(stf.StructType(fields), cols, rows)
dev = g.batch_to_device(batch)
os.makedirs("/dev/shm/huge", exist_ok=True)
path = "/dev/shm/huge/t.tfrecord"

def timed(name, fn, reps=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps): out = fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / reps
    print(f"{name:28s} {dt*1000:8.2f} ms", flush=True)
    return out

timed("write_batch_to_file", lambda: g.write_batch_to_file(dev, path, "Example"))
size = os.path.getsize(path)
print(f"file = {size/1e6:.1f} MB  avg rec = {size//rows} B")
timed("encode_device only", lambda: g.encode_device(dev, "Example"))
timed("read+decode pipelined", lambda: g.read_file_to_batch_pipelined(
    path, batch.schema, "Example", True))
data = timed("read_file_to_device", lambda: g.read_file_to_device(path))
off, lens = timed("scan_frames_device", lambda: g.scan_frames_device(data))
timed("decode_device (crc on)", lambda: g.decode_device(
    data, off, lens, batch.schema, "Example", True))
timed("decode_device (crc off)", lambda: g.decode_device(
    data, off, lens, batch.schema, "Example", False))
