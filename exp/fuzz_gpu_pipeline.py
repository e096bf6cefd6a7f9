"""GPU pipeline fuzz: random batch shapes and slice sizes through the sliced
write/read paths, group decode, and partition gather — outputs must match
the host codec exactly."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import spark_tfrecord_amd as stf
from spark_tfrecord_amd.columnar import RecordBatch, column_from_values
from spark_tfrecord_amd.engine import cpu as cpu_engine
from spark_tfrecord_amd.engine import gpu as g

def rand_batch(rng, n):
    schema = stf.StructType([
        stf.StructField("i", stf.LongType(), True),
        stf.StructField("a", stf.ArrayType(stf.LongType()), True),
        stf.StructField("f", stf.ArrayType(stf.FloatType()), True),
        stf.StructField("s", stf.ArrayType(stf.StringType()), True),
    ])
    cols = [
        column_from_values([int(v) if v % 5 else None
                            for v in rng.integers(-2**62, 2**62, n)],
                           stf.LongType(), True, "i"),
        column_from_values([list(rng.integers(-99, 99, int(k % 7)))
                            for k in rng.integers(0, 100, n)],
                           stf.ArrayType(stf.LongType()), True, "a"),
        column_from_values([list(rng.random(int(k % 5)).astype(float))
                            for k in rng.integers(0, 100, n)],
                           stf.ArrayType(stf.FloatType()), True, "f"),
        column_from_values([[("x" * int(rng.integers(0, 30)))
                             for _ in range(int(k % 4))]
                            for k in rng.integers(0, 100, n)],
                           stf.ArrayType(stf.StringType()), True, "s"),
    ]
    return RecordBatch(schema, cols, n)

def eq(a, b):
    assert a.num_rows == b.num_rows
    for ca, cb in zip(a.columns, b.columns):
        for attr in ("presence", "row_off", "values", "elem_off", "list_off", "sub_off"):
            va, vb = getattr(ca, attr), getattr(cb, attr)
            assert (va is None) == (vb is None), attr
            if va is not None:
                np.testing.assert_array_equal(np.asarray(va), np.asarray(vb))

rng = np.random.default_rng(int(os.environ.get("FUZZ_SEED", 0)))
base = "/dev/shm/gpu_fuzz"
os.makedirs(base, exist_ok=True)
trials = int(os.environ.get("FUZZ_TRIALS", 25))
for t in range(trials):
    n = int(rng.integers(1, 200_000))
    batch = rand_batch(rng, n)
    img_cpu = cpu_engine.encode_batch(batch, "Example")
    path = f"{base}/f{t % 3}.tfrecord"
    slices = int(rng.integers(1, 9))
    g._READ_SLICE = int(rng.integers(1, 200)) << 12  # 4KB..800KB slices
    nbytes = g.write_batch_to_file(g.batch_to_device(batch), path, "Example",
                                   slices=slices)
    assert open(path, "rb").read() == img_cpu, f"trial {t}: file bytes differ"
    out = g.batch_to_host(g.read_file_to_batch_pipelined(
        path, batch.schema, "Example", verify_crc=True))
    eq(batch, out)
    # group decode of 2 files
    path2 = f"{base}/g{t % 3}.tfrecord"
    n2 = int(rng.integers(1, 5000))
    batch2 = rand_batch(rng, n2)
    g.write_batch_to_file(g.batch_to_device(batch2), path2, "Example")
    gb, counts = g.read_files_to_batch([path, path2], batch.schema, "Example")
    assert list(counts) == [n, n2], (counts, n, n2)
    # partition gather
    P = int(rng.integers(1, 12))
    codes = rng.integers(0, P, n).astype(np.int64)
    img, ranges = g.encode_partitions_device(g.batch_to_device(batch), codes,
                                             P, "Example")
    total = sum(hi - lo for _, lo, hi in ranges)
    assert total == img.numel() == len(img_cpu)
    if t % 5 == 4:
        print(f"trial {t+1}/{trials} ok (n={n}, slices={slices}, rs={g._READ_SLICE})")
print("gpu pipeline fuzz: all ok")
