"""Edge-shape battery: schema widths at the engine's limits, degenerate row
counts, many files, large single records — shapes that break offset math
first (mirrors the reference suite's 15-column wide-schema test at
TFRecordIOSuite.scala:117-138, pushed further)."""

import os

import numpy as np
import pytest

import spark_tfrecord_amd as stf
from spark_tfrecord_amd.columnar import RecordBatch, column_from_values
from spark_tfrecord_amd.engine import cpu as cpu_engine


def roundtrip(batch, record_type="Example"):
    img = cpu_engine.encode_batch(batch, record_type)
    out = cpu_engine.decode_buffer(np.frombuffer(img, np.uint8), batch.schema,
                                   record_type)
    for ca, cb in zip(batch.columns, out.columns):
        np.testing.assert_array_equal(np.asarray(ca.presence), np.asarray(cb.presence))
        np.testing.assert_array_equal(np.asarray(ca.row_off), np.asarray(cb.row_off))
        np.testing.assert_array_equal(np.asarray(ca.values), np.asarray(cb.values))
    return out


class TestWideSchema:
    def test_64_fields_roundtrip(self):
        n = 50
        fields, cols = [], []
        for i in range(64):
            name = f"f{i:02d}"
            fields.append(stf.StructField(name, stf.LongType(), True))
            cols.append(column_from_values(
                list(range(i, i + n)), stf.LongType(), True, name))
        batch = RecordBatch(stf.StructType(fields), cols, n)
        roundtrip(batch)

    def test_65_fields_roundtrip(self):
        # no field-count limit on either engine (device column table is
        # sized by the schema)
        n = 5
        fields, cols = [], []
        for i in range(65):
            name = f"g{i:02d}"
            fields.append(stf.StructField(name, stf.FloatType(), True))
            cols.append(column_from_values(
                [float(i)] * n, stf.FloatType(), True, name))
        batch = RecordBatch(stf.StructType(fields), cols, n)
        roundtrip(batch)

    def test_long_feature_names(self):
        name = "n" * 500
        batch = RecordBatch(
            stf.StructType([stf.StructField(name, stf.LongType(), True)]),
            [column_from_values([1, 2, 3], stf.LongType(), True, name)], 3)
        roundtrip(batch)


class TestDegenerateShapes:
    def test_zero_rows(self):
        batch = RecordBatch(
            stf.StructType([stf.StructField("x", stf.LongType(), True)]),
            [column_from_values([], stf.LongType(), True, "x")], 0)
        img = cpu_engine.encode_batch(batch, "Example")
        assert img == b""

    def test_single_huge_record(self):
        vals = list(range(200_000))  # ~1 MB varint payload in one record
        dt = stf.ArrayType(stf.LongType())
        batch = RecordBatch(
            stf.StructType([stf.StructField("big", dt, True)]),
            [column_from_values([vals], dt, True, "big")], 1)
        out = roundtrip(batch)
        assert len(np.asarray(out.columns[0].values)) == 200_000

    def test_alternating_presence(self):
        n = 1001
        vals = [i if i % 2 else None for i in range(n)]
        batch = RecordBatch(
            stf.StructType([stf.StructField("x", stf.LongType(), True)]),
            [column_from_values(vals, stf.LongType(), True, "x")], n)
        roundtrip(batch)

    def test_extreme_varints(self):
        vals = [0, 1, -1, 2**63 - 1, -(2**63), 127, 128, 2**32, -(2**32)]
        batch = RecordBatch(
            stf.StructType([stf.StructField("x", stf.LongType(), True)]),
            [column_from_values(vals, stf.LongType(), True, "x")], len(vals))
        roundtrip(batch)

    def test_empty_strings_and_lists(self):
        dt = stf.ArrayType(stf.StringType())
        vals = [[], [""], ["", "a", ""], None, ["bb"]]
        batch = RecordBatch(
            stf.StructType([stf.StructField("s", dt, True)]),
            [column_from_values(vals, dt, True, "s")], len(vals))
        roundtrip(batch)


class TestManyFiles:
    def test_hundred_shards_roundtrip(self, tmp_sandbox):
        out = str(tmp_sandbox / "many")
        n = 1000
        stf.write_tfrecord({"x": np.arange(n, dtype=np.int64)}, out,
                           engine="cpu", num_shards=100)
        parts = [f for f in os.listdir(out) if f.startswith("part-")]
        assert len(parts) == 100
        df = stf.read_tfrecord(out, engine="cpu").sort("x")
        assert df.count() == n
        assert [r["x"] for r in df.collect()[:3]] == [0, 1, 2]

    def test_some_empty_shards(self, tmp_sandbox):
        out = str(tmp_sandbox / "empties")
        os.makedirs(out)
        # hand-build: 2 real shards + 2 empty files
        stf.write_tfrecord({"x": np.arange(10, dtype=np.int64)}, out,
                           engine="cpu", mode="append", num_shards=2)
        open(os.path.join(out, "part-90000-empty.tfrecord"), "wb").close()
        open(os.path.join(out, "part-90001-empty.tfrecord"), "wb").close()
        df = stf.read_tfrecord(out, engine="cpu")
        assert df.count() == 10
