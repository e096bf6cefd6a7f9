"""Extended GPU-vs-host equality campaign: random batches of every record
type through encode+decode, GPU output must equal the host codec exactly."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import pyarrow as pa
import spark_tfrecord_amd as stf
from spark_tfrecord_amd.columnar import RecordBatch, column_from_values
from spark_tfrecord_amd.engine import cpu as cpu_engine
from spark_tfrecord_amd.engine import gpu as g
from spark_tfrecord_amd.arrow_interop import table_to_batch
from spark_tfrecord_amd.infer import byte_array_schema

def eq(a, b):
    for ca, cb in zip(a.columns, b.columns):
        for attr in ("presence", "row_off", "values", "elem_off", "list_off", "sub_off"):
            va, vb = getattr(ca, attr), getattr(cb, attr)
            assert (va is None) == (vb is None)
            if va is not None:
                np.testing.assert_array_equal(np.asarray(va), np.asarray(vb))

rng = np.random.default_rng(0)
for t in range(50):
    kind = t % 3
    n = int(rng.integers(1, 30_000))
    if kind == 0:  # Example, mixed
        schema = stf.StructType([
            stf.StructField("a", stf.LongType(), True),
            stf.StructField("b", stf.ArrayType(stf.FloatType()), True),
            stf.StructField("c", stf.StringType(), True)])
        cols = [column_from_values([int(v) if v % 3 else None for v in rng.integers(0, 9, n)], stf.LongType(), True, "a"),
                column_from_values([list(rng.random(int(k % 9)).astype(float)) for k in rng.integers(0, 100, n)], stf.ArrayType(stf.FloatType()), True, "b"),
                column_from_values([("q" * int(rng.integers(0, 40))) if v % 4 else None for v in rng.integers(0, 9, n)], stf.StringType(), True, "c")]
        batch, rt = RecordBatch(schema, cols, n), "Example"
    elif kind == 1:  # SequenceExample ragged
        dt = stf.ArrayType(stf.ArrayType(stf.LongType()))
        schema = stf.StructType([
            stf.StructField("c", stf.FloatType(), True),
            stf.StructField("r", dt, True)])
        rag = [[list(rng.integers(0, 99, int(rng.integers(0, 5)))) for _ in range(int(rng.integers(0, 4)))] for _ in range(n)]
        cols = [column_from_values(rng.random(n).astype(np.float32), stf.FloatType(), True, "c"),
                column_from_values(rag, dt, True, "r")]
        batch, rt = RecordBatch(schema, cols, n), "SequenceExample"
    else:  # ByteArray
        payloads = [rng.bytes(int(rng.integers(0, 400))) for _ in range(n)]
        table = pa.table({"byteArray": pa.array(payloads, type=pa.large_binary())})
        batch, rt = table_to_batch(table, byte_array_schema()), "ByteArray"
        schema = byte_array_schema()
    cpu_img = cpu_engine.encode_batch(batch, rt)
    gpu_img = g.encode_batch_from_cpu(batch, rt)
    assert cpu_img == gpu_img, f"trial {t} ({rt}, n={n}): encode differs"
    data = np.frombuffer(cpu_img, np.uint8)
    eq(cpu_engine.decode_buffer(data, schema, rt),
       g.decode_buffer_to_cpu(data, schema, rt))
    if t % 10 == 9:
        print(f"trial {t+1}/50 ok ({rt}, n={n})")
print("gpu equality campaign: all ok")
