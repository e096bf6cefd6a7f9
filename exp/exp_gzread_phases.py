"""Phase breakdown of the config-5 read (gzip ByteArray, 32 shards)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import pyarrow as pa
import torch
import spark_tfrecord_amd as stf
from spark_tfrecord_amd.io import paths as P
from spark_tfrecord_amd.engine import gpu as g
from spark_tfrecord_amd.infer import byte_array_schema
from spark_tfrecord_amd.arrow_interop import batch_to_table

rows = 1_000_000
rng = np.random.default_rng(9)
payloads = [rng.bytes(200) for _ in range(rows)]
t = pa.table({"byteArray": pa.array(payloads, type=pa.large_binary())})
d = "/dev/shm/gzp/t"
stf.write_tfrecord(t, d, record_type="ByteArray", codec="gzip",
                   mode="overwrite", engine="cpu", num_shards=32)
files = P.list_data_files(d)
schema = byte_array_schema()

def timed(name, fn, reps=3):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        r = fn()
    torch.cuda.synchronize()
    print(f"{name:30s} {(time.perf_counter()-t0)/reps*1000:8.2f} ms", flush=True)
    return r

timed("parse headers x32", lambda: [P.parse_gz_segments_file(f) for f in files])
timed("gz_device_meta x32", lambda: [g.gz_device_meta(f) for f in files])

metas = [P.parse_gz_segments_file(f) for f in files]
sizes = [sum(u for _, u in m[1]) for m in metas]
bounds = np.zeros(33, np.int64); np.cumsum(sizes, out=bounds[1:])

def inflate_only():
    data = torch.empty(int(bounds[-1]), dtype=torch.uint8, device="cuda")
    ok = g._device_inflate_group(
        data, [(f, m, int(b)) for f, m, b in zip(files, metas, bounds[:-1])],
        torch.device("cuda"))
    assert ok
    return data

data = timed("inflate group (32 files)", inflate_only)
off, lens = timed("scan_frames", lambda: g.scan_frames_device(data))
batch = timed("decode ByteArray (crc on)", lambda: g.decode_device(
    data, off, lens, schema, "ByteArray", True))
host = timed("batch_to_host", lambda: g.batch_to_host(batch))
timed("batch_to_table", lambda: batch_to_table(host))
timed("FULL read_tfrecord", lambda: stf.read_tfrecord(
    d, record_type="ByteArray", engine="gpu"), reps=3)
