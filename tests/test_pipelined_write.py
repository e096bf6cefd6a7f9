"""Pipelined single-shard write path.

The default (num_shards=1, codec=None) API write streams the table in row
chunks: host conversion and encode of chunk k+1 overlap the file append of
chunk k, with the temp + atomic-rename commit guarantee preserved (the
reference gets the same guarantee from Spark's task-commit protocol,
TFRecordOutputWriter.scala:40-43 + FileFormatWriter). TFRecord frames are
freely concatenable, so the chunked file must be byte-identical in content
to the one-shot file.
"""

import os

import numpy as np
import pyarrow as pa
import pytest

import spark_tfrecord_amd as stf
from spark_tfrecord_amd.io import paths as P
from spark_tfrecord_amd.io import writer as W


def _table(rows, seed=0):
    rng = np.random.default_rng(seed)
    return pa.table({
        "id": np.arange(rows, dtype=np.int64),
        "x": rng.random(rows).astype(np.float32),
        "s": pa.array([f"r{i}" for i in range(rows)]),
    })


def _read_ids(path):
    t = stf.read_tfrecord(path, engine="cpu").to_arrow_table()
    return sorted(t.column("id").to_pylist())


class TestPipelinedSingleShard:
    def test_chunked_matches_oneshot(self, tmp_path, monkeypatch):
        rows = 5_000
        monkeypatch.setattr(W, "_PIPE_CHUNK_ROWS", 700)  # force many chunks
        a = str(tmp_path / "chunked")
        stf.write_tfrecord(_table(rows), a, engine="cpu", job_id="jobx")
        monkeypatch.setattr(W, "_PIPE_CHUNK_ROWS", 10**9)  # one-shot path
        b = str(tmp_path / "oneshot")
        stf.write_tfrecord(_table(rows), b, engine="cpu", job_id="jobx")
        fa = [f for f in os.listdir(a) if f != "_SUCCESS"]
        fb = [f for f in os.listdir(b) if f != "_SUCCESS"]
        assert fa == fb  # identical part-file naming (same job_id)
        with open(os.path.join(a, fa[0]), "rb") as f1, \
                open(os.path.join(b, fb[0]), "rb") as f2:
            assert f1.read() == f2.read()

    def test_roundtrip_through_pipeline(self, tmp_path, monkeypatch):
        monkeypatch.setattr(W, "_PIPE_CHUNK_ROWS", 333)
        out = str(tmp_path / "ds")
        stf.write_tfrecord(_table(2_000, seed=3), out, engine="cpu")
        assert _read_ids(out) == list(range(2_000))

    def test_uneven_tail_chunk(self, tmp_path, monkeypatch):
        monkeypatch.setattr(W, "_PIPE_CHUNK_ROWS", 1000)
        out = str(tmp_path / "ds")
        stf.write_tfrecord(_table(2_501), out, engine="cpu")
        assert _read_ids(out) == list(range(2_501))

    def test_failure_leaves_no_visible_file(self, tmp_path, monkeypatch):
        monkeypatch.setattr(W, "_PIPE_CHUNK_ROWS", 100)
        out = str(tmp_path / "ds")
        t = pa.table({"id": np.arange(500, dtype=np.int64),
                      "x": pa.array([None] * 500, type=pa.float32())})
        schema = stf.StructType([
            stf.StructField("id", stf.LongType(), True),
            stf.StructField("x", stf.FloatType(), False),  # non-nullable
        ])
        with pytest.raises(ValueError, match="non-nullable"):
            stf.write_tfrecord(t, out, schema=schema, engine="cpu")
        # neither a data file nor a temp leftover is listed as data
        assert P.list_data_files(out) == []

    def test_gzip_stays_on_oneshot_path(self, tmp_path, monkeypatch):
        # codec writes must produce one coherent gzip member stream
        monkeypatch.setattr(W, "_PIPE_CHUNK_ROWS", 100)
        out = str(tmp_path / "gz")
        stf.write_tfrecord(_table(1_000), out, codec="gzip", engine="cpu")
        assert _read_ids(out) == list(range(1_000))
