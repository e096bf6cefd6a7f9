"""Round-trip throughput across record shapes/sizes (write+read+decode,
CRC verified), 1 GPU."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import spark_tfrecord_amd as stf
from spark_tfrecord_amd.columnar import RecordBatch, WireColumn
from spark_tfrecord_amd.schema import KIND_BYTES, KIND_FLOAT, KIND_INT64
from spark_tfrecord_amd.engine import gpu as g

def mk(rows, ints, floats, strbytes):
    rng = np.random.default_rng(0)
    fields = [stf.StructField("id", stf.LongType(), True)]
    cols = [WireColumn(KIND_INT64, False, np.ones(rows, np.uint8),
                       np.arange(rows + 1, dtype=np.int64),
                       rng.integers(0, 2**62, rows).astype(np.int64))]
    if ints:
        fields.append(stf.StructField("ints", stf.ArrayType(stf.LongType()), True))
        cols.append(WireColumn(KIND_INT64, False, np.ones(rows, np.uint8),
                               np.arange(0, (rows+1)*ints, ints, dtype=np.int64),
                               rng.integers(0, 2**31, rows*ints).astype(np.int64)))
    if floats:
        fields.append(stf.StructField("floats", stf.ArrayType(stf.FloatType()), True))
        cols.append(WireColumn(KIND_FLOAT, False, np.ones(rows, np.uint8),
                               np.arange(0, (rows+1)*floats, floats, dtype=np.int64),
                               rng.random(rows*floats).astype(np.float32)))
    if strbytes:
        fields.append(stf.StructField("blob", stf.StringType(), True))
        data = rng.integers(65, 90, rows*strbytes).astype(np.uint8)
        cols.append(WireColumn(KIND_BYTES, False, np.ones(rows, np.uint8),
                               np.arange(rows + 1, dtype=np.int64), data,
                               elem_off=np.arange(0, (rows+1)*strbytes, strbytes, dtype=np.int64)))
    return RecordBatch(stf.StructType(fields), cols, rows)

os.makedirs("/dev/shm/shapes", exist_ok=True)
for name, rows, ints, floats, sb in [
        ("tiny 20B/rec", 4_000_000, 0, 0, 0),
        ("small 60B/rec", 2_000_000, 4, 4, 0),
        ("flagship 215B/rec", 1_000_000, 8, 16, 24),
        ("large 1.1KB/rec", 200_000, 16, 64, 512),
        ("huge 16KB/rec", 16_000, 0, 256, 15_000)]:
    batch = mk(rows, ints, floats, sb)
    dev = g.batch_to_device(batch)
    path = "/dev/shm/shapes/t.tfrecord"
    def step():
        g.write_batch_to_file(dev, path, "Example")
        return g.read_file_to_batch_pipelined(path, batch.schema, "Example", True)
    step(); step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5): step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 5
    size = os.path.getsize(path)
    print(f"{name:22s} rows={rows:>9,} file={size/1e6:8.1f}MB  "
          f"{rows/dt/1e6:7.1f}M rows/s  {size/dt/1e9*2:6.1f} GB/s rt")
