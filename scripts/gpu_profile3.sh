set -e
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out/prof5
export TMPDIR=/tmp
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof5" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 5 --warmup 2 > "$GRAFT_REPO_ROOT/gpurun_out/prof5.log" 2>&1
cd "$GRAFT_REPO_ROOT"
tail -1 gpurun_out/prof5.log
timeout 420 python - <<'PY' 2>&1 | tail -3
import time
import numpy as np
import pyarrow as pa
import spark_tfrecord_amd as stf
rows = 1_000_000
rng = np.random.default_rng(3)
lens1 = rng.integers(1, 4, rows)
rag = pa.array([[[float(j) for j in range(int(k))]] for k in lens1],
               type=pa.large_list(pa.large_list(pa.float32())))
t = pa.table({"sid": pa.array(np.arange(rows, dtype=np.int64)), "rag": rag})
schema = stf.StructType([
    stf.StructField("sid", stf.LongType(), True),
    stf.StructField("rag", stf.ArrayType(stf.ArrayType(stf.FloatType())), True)])
out = "/dev/shm/seq_api/t"
stf.write_tfrecord(t, out, record_type="SequenceExample", schema=schema, mode="overwrite", engine="gpu")
t0=time.perf_counter()
stf.write_tfrecord(t, out, record_type="SequenceExample", schema=schema, mode="overwrite", engine="gpu")
t1=time.perf_counter()
df = stf.read_tfrecord(out, record_type="SequenceExample", engine="gpu")
t2=time.perf_counter()
assert df.count() == rows
print(f"SequenceExample API warm: write {rows/(t1-t0)/1e6:.1f}M rows/s, read {rows/(t2-t1)/1e6:.1f}M rows/s")
PY
