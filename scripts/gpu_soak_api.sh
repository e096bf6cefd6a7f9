set -e
cd "$GRAFT_REPO_ROOT"
timeout 420 python -m pytest tests -m gpu -x -q 2>&1 | tail -1
timeout 300 python bench.py --steps 10 --warmup 3 2>&1 | tail -1 | python3 -c "import json,sys; d=json.load(sys.stdin); print('flagship:', round(d['value']/1e6,1), 'M rows/s,', round(d['ms_per_step'],2), 'ms/step')"
timeout 420 python - <<'PY' 2>&1 | tail -6
import time
import numpy as np
import pyarrow as pa
import spark_tfrecord_amd as stf

# SequenceExample 1M rows through the full API on GPU
rows = 1_000_000
rng = np.random.default_rng(3)
lens1 = rng.integers(1, 4, rows)
rag = pa.array([[[float(j) for j in range(int(k))]] for k in lens1])
t = pa.table({"sid": pa.array(np.arange(rows, dtype=np.int64)), "rag": rag})
schema = stf.StructType([
    stf.StructField("sid", stf.LongType(), True),
    stf.StructField("rag", stf.ArrayType(stf.ArrayType(stf.FloatType())), True)])
out = "/dev/shm/seq_api/t"
t0=time.perf_counter()
stf.write_tfrecord(t, out, record_type="SequenceExample", schema=schema,
                   mode="overwrite", engine="gpu")
t1=time.perf_counter()
df = stf.read_tfrecord(out, record_type="SequenceExample", engine="gpu")
t2=time.perf_counter()
assert df.count() == rows
print(f"SequenceExample API: write {rows/(t1-t0)/1e6:.1f}M rows/s, read {rows/(t2-t1)/1e6:.1f}M rows/s")

# gzip WRITE via gpu engine + read back
out2 = "/dev/shm/seq_api/gz"
d2 = pa.table({"x": pa.array(np.arange(200_000, dtype=np.int64))})
stf.write_tfrecord(d2, out2, codec="gzip", mode="overwrite", engine="gpu", num_shards=4)
df2 = stf.read_tfrecord(out2, engine="gpu")
assert df2.count() == 200_000
import os
assert all(f.endswith(".gz") or f.startswith("_") for f in os.listdir(out2))
print("gzip write via GPU engine: ok")
PY
