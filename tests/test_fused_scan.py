"""Differential tests: the fused cursor scan (single pass, shared loads for
parse + CRC) must produce byte-identical stats and CRCs to the two-pass
reference on arbitrary inputs (csrc/codec_core.h ScanCur)."""

import numpy as np
import pytest

import spark_tfrecord_amd as stf
from spark_tfrecord_amd import _native
from spark_tfrecord_amd.columnar import RecordBatch, column_from_values, schema_blob
from spark_tfrecord_amd.engine import cpu as cpu_engine


def both(img: bytes, schema, record_type="Example"):
    data = np.frombuffer(img, np.uint8)
    off, lens = _native.scan_frames(data, False)
    fmt = _native.FMT_SEQUENCE if record_type == "SequenceExample" else _native.FMT_EXAMPLE
    blob = schema_blob(schema)
    plain = _native.scan_stats_debug(data, off, lens, fmt, blob, False)
    fused = _native.scan_stats_debug(data, off, lens, fmt, blob, True)
    return plain, fused


def assert_same(plain, fused):
    np.testing.assert_array_equal(plain["rc"], fused["rc"])
    np.testing.assert_array_equal(plain["stats"], fused["stats"])
    np.testing.assert_array_equal(plain["crc"], fused["crc"])


class TestFusedScan:
    @pytest.mark.parametrize("seed", range(6))
    def test_random_example_batches(self, seed):
        rng = np.random.default_rng(seed)
        n = int(rng.integers(1, 300))
        schema = stf.StructType([
            stf.StructField("i", stf.LongType(), True),
            stf.StructField("f", stf.FloatType(), True),
            stf.StructField("s", stf.StringType(), True),
            stf.StructField("a", stf.ArrayType(stf.LongType()), True),
        ])
        cols = [
            column_from_values(
                [int(v) if v % 3 else None for v in rng.integers(-2**62, 2**62, n)],
                stf.LongType(), True, "i"),
            column_from_values(rng.random(n).astype(np.float32),
                               stf.FloatType(), True, "f"),
            column_from_values([f"s{v}" * (v % 4) for v in range(n)],
                               stf.StringType(), True, "s"),
            column_from_values([list(rng.integers(-5, 5, v % 7)) for v in range(n)],
                               stf.ArrayType(stf.LongType()), True, "a"),
        ]
        batch = RecordBatch(schema, cols, n)
        plain, fused = both(cpu_engine.encode_batch(batch, "Example"), schema)
        assert_same(plain, fused)

    def test_sequence_example(self):
        rng = np.random.default_rng(9)
        n = 120
        dt = stf.ArrayType(stf.ArrayType(stf.FloatType()))
        schema = stf.StructType([
            stf.StructField("c", stf.LongType(), True),
            stf.StructField("r", dt, True),
        ])
        rag = [[list(rng.random(rng.integers(0, 5)).astype(float))
                for _ in range(rng.integers(0, 4))] for _ in range(n)]
        cols = [column_from_values(list(range(n)), stf.LongType(), True, "c"),
                column_from_values(rag, dt, True, "r")]
        batch = RecordBatch(schema, cols, n)
        plain, fused = both(cpu_engine.encode_batch(batch, "SequenceExample"),
                            schema, "SequenceExample")
        assert_same(plain, fused)

    def test_corrupted_bytes_same_verdict(self):
        rng = np.random.default_rng(3)
        n = 40
        schema = stf.StructType([stf.StructField("x", stf.ArrayType(stf.LongType()),
                                                 True)])
        cols = [column_from_values([list(rng.integers(0, 99, 5))] * n,
                                   stf.ArrayType(stf.LongType()), True, "x")]
        img = bytearray(cpu_engine.encode_batch(RecordBatch(schema, cols, n),
                                                "Example"))
        # flip payload bytes in a few records (skip the frame headers)
        for pos in (20, 150, 400):
            if pos < len(img):
                img[pos] ^= 0x5A
        data = np.frombuffer(bytes(img), np.uint8)
        off, lens = _native.scan_frames(data, False)
        blob = schema_blob(schema)
        plain = _native.scan_stats_debug(data, off, lens, _native.FMT_EXAMPLE,
                                         blob, False)
        fused = _native.scan_stats_debug(data, off, lens, _native.FMT_EXAMPLE,
                                         blob, True)
        # both forms must agree on which records parse (exact error codes may
        # differ on malformed bytes — both still reject) and on CRCs
        np.testing.assert_array_equal(plain["rc"] == 0, fused["rc"] == 0)
        np.testing.assert_array_equal(plain["crc"], fused["crc"])

    def test_unknown_features_and_long_names(self):
        # schema only knows one of the features; unknown ones are skipped by
        # the scan but still CRC'd by the fused cursor
        name = "k" * 100
        schema_w = stf.StructType([
            stf.StructField(name, stf.LongType(), True),
            stf.StructField("other", stf.StringType(), True),
        ])
        cols = [column_from_values([1, 2, 3], stf.LongType(), True, name),
                column_from_values(["a", "bb", None], stf.StringType(), True,
                                   "other")]
        img = cpu_engine.encode_batch(RecordBatch(schema_w, cols, 3), "Example")
        schema_narrow = stf.StructType([stf.StructField(name, stf.LongType(), True)])
        plain, fused = both(img, schema_narrow)
        assert_same(plain, fused)
