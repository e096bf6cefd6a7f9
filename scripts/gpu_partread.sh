cd "$GRAFT_REPO_ROOT"
timeout 420 python - <<'PY' 2>&1 | tail -8
import os, time
import numpy as np
import pyarrow as pa
import spark_tfrecord_amd as stf

rows = 2_000_000
rng = np.random.default_rng(0)
table = pa.table({
    "date": pa.array([f"2026-09-{d:02d}" for d in rng.integers(1, 11, rows)]),
    "uid": pa.array(rng.integers(0, 2**62, rows)),
    "score": pa.array(rng.random(rows).astype(np.float32)),
    "feats": pa.array(list(rng.random((rows, 8)).astype(np.float32))),
})
out = "/dev/shm/partread/t"
t0=time.perf_counter()
stf.write_tfrecord(table, out, partition_by=["date"], mode="overwrite", engine="gpu")
t1=time.perf_counter()
df = stf.read_tfrecord(out, engine="gpu")
t2=time.perf_counter()
assert df.count() == rows, df.count()
assert sorted(df.columns) == ["date", "feats", "score", "uid"]
# verify one partition's contents
s = df.to_arrow_table().group_by("date").aggregate([("uid", "count")])
print("partitions:", s.num_rows)
print(f"write {t1-t0:.2f}s ({rows/(t1-t0)/1e6:.1f}M rows/s)  read {t2-t1:.2f}s ({rows/(t2-t1)/1e6:.1f}M rows/s)")
rep = stf.validate_tfrecord(out)
print("validate:", rep)
PY
