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


class TestSlicedConversion:
    """arrow_to_wire on sliced (offset != 0) chunks must take the rebased
    contiguous fast paths, not the gather/to_pylist fallbacks, and must
    produce values identical to converting the whole table."""

    def _roundtrip_eq(self, tab, schema=None):
        from spark_tfrecord_amd.arrow_interop import (schema_from_arrow,
                                                      table_to_batch)
        from spark_tfrecord_amd.engine import cpu as cpu_engine
        sch = schema or schema_from_arrow(tab.schema)
        whole = cpu_engine.encode_batch(table_to_batch(tab, sch), "Example")
        parts = []
        step = max(1, tab.num_rows // 3)
        for lo in range(0, tab.num_rows, step):
            ch = tab.slice(lo, min(step, tab.num_rows - lo))
            parts.append(cpu_engine.encode_batch(table_to_batch(ch, sch),
                                                 "Example"))
        assert b"".join(parts) == bytes(whole)

    def test_numeric_scalar_and_list(self):
        rng = np.random.default_rng(0)
        off = np.concatenate([[0], np.cumsum(rng.integers(0, 5, 100))])
        vals = rng.integers(-9, 9, off[-1]).astype(np.int64)
        tab = pa.table({
            "a": np.arange(100, dtype=np.int64),
            "f": rng.random(100).astype(np.float32),
            "l": pa.LargeListArray.from_arrays(off.astype(np.int64), vals),
        })
        self._roundtrip_eq(tab)

    def test_string_scalar_and_list(self):
        rng = np.random.default_rng(1)
        strs = [f"v{i}" * (i % 4) for i in range(100)]
        off = np.concatenate([[0], np.cumsum(rng.integers(0, 4, 100))])
        toks = pa.array([f"t{i%13}" for i in range(off[-1])])
        tab = pa.table({
            "s": pa.array(strs),
            "ts": pa.LargeListArray.from_arrays(off.astype(np.int64),
                                                toks.cast(pa.large_string())),
        })
        self._roundtrip_eq(tab)

    def test_nested_2d_sequence(self):
        from spark_tfrecord_amd.arrow_interop import table_to_batch
        rng = np.random.default_rng(2)
        o2 = np.concatenate([[0], np.cumsum(rng.integers(0, 4, 50))])
        inner = pa.LargeListArray.from_arrays(
            o2.astype(np.int64), rng.random(o2[-1]).astype(np.float32))
        o1 = np.concatenate([[0], np.cumsum(rng.integers(0, 3, 20))])
        assert o1[-1] <= 50
        outer = pa.LargeListArray.from_arrays(
            o1.astype(np.int64), inner.slice(0, int(o1[-1])))
        tab = pa.table({"seq": outer})
        sch = stf.StructType([stf.StructField(
            "seq", stf.ArrayType(stf.ArrayType(stf.FloatType())), True)])
        from spark_tfrecord_amd.engine import cpu as cpu_engine
        whole = cpu_engine.encode_batch(table_to_batch(tab, sch),
                                        "SequenceExample")
        parts = []
        for lo in range(0, 20, 7):
            ch = tab.slice(lo, min(7, 20 - lo))
            parts.append(cpu_engine.encode_batch(table_to_batch(ch, sch),
                                                 "SequenceExample"))
        assert b"".join(parts) == bytes(whole)

    def test_sliced_with_nulls(self):
        tab = pa.table({
            "x": pa.array([1, None, 3, None, 5, 6, None, 8], type=pa.int64()),
            "s": pa.array(["a", None, "c", "d", None, "f", "g", "h"]),
        })
        self._roundtrip_eq(tab)

    def test_empty_slice_of_offset_column(self):
        """An empty slice taken at a non-zero offset must still produce
        wire columns whose offsets start at 0."""
        from spark_tfrecord_amd.arrow_interop import table_to_batch, schema_from_arrow
        rng = np.random.default_rng(4)
        off = np.concatenate([[0], np.cumsum(rng.integers(1, 4, 50))])
        tab = pa.table({
            "s": pa.array([f"x{i}" for i in range(50)]),
            "l": pa.LargeListArray.from_arrays(
                off.astype(np.int64), rng.integers(0, 9, off[-1]).astype(np.int64)),
        })
        empty = tab.slice(30, 0)
        b = table_to_batch(empty, schema_from_arrow(tab.schema))
        for c in b.columns:
            assert c.row_off[0] == 0
            assert len(c.values) == 0
            if c.elem_off is not None:
                assert c.elem_off[0] == 0


class TestMultiChunkInput:
    def test_concat_table_multi_chunk_write(self, tmp_path):
        """pa.concat_tables yields multi-chunk columns; the conversion must
        concatenate (combine_chunks fallback) and round-trip exactly."""
        rng = np.random.default_rng(9)
        parts = []
        for k in range(4):
            parts.append(pa.table({
                "id": np.arange(k * 1000, (k + 1) * 1000, dtype=np.int64),
                "l": pa.LargeListArray.from_arrays(
                    np.arange(0, 2002, 2, dtype=np.int64),
                    rng.integers(0, 99, 2000).astype(np.int64)),
                "s": pa.array([f"c{k}_{i}" for i in range(1000)]),
            }))
        tab = pa.concat_tables(parts)
        assert tab.column("id").num_chunks == 4
        out = str(tmp_path / "mc")
        stf.write_tfrecord(tab, out, engine="cpu")
        got = stf.read_tfrecord(out, engine="cpu").to_arrow_table()
        assert sorted(got.column("id").to_pylist()) == list(range(4000))
        by_id = {r["id"]: r for r in got.to_pylist()}
        want = tab.to_pylist()
        for w in want[:50] + want[-50:]:
            assert by_id[w["id"]]["l"] == w["l"]
            assert by_id[w["id"]]["s"] == w["s"]
