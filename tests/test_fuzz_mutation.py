"""Mutation fuzzing: the native parsers index raw byte buffers, so hostile
input must fail with Python exceptions — never a crash or silent corruption.
Every mutation is also run through BOTH scan forms (fused cursor vs
two-pass), which must agree on per-record verdicts."""

import numpy as np
import pytest

import spark_tfrecord_amd as stf
from spark_tfrecord_amd import _native
from spark_tfrecord_amd.columnar import RecordBatch, column_from_values, schema_blob
from spark_tfrecord_amd.engine import cpu as cpu_engine


def _image(seed=0, n=60):
    rng = np.random.default_rng(seed)
    schema = stf.StructType([
        stf.StructField("i", stf.LongType(), True),
        stf.StructField("a", stf.ArrayType(stf.LongType()), True),
        stf.StructField("s", stf.ArrayType(stf.StringType()), True),
    ])
    cols = [
        column_from_values(rng.integers(-2**60, 2**60, n), stf.LongType(), True, "i"),
        column_from_values([list(rng.integers(0, 999, k % 6)) for k in range(n)],
                           stf.ArrayType(stf.LongType()), True, "a"),
        column_from_values([[f"s{j}" for j in range(k % 4)] for k in range(n)],
                           stf.ArrayType(stf.StringType()), True, "s"),
    ]
    batch = RecordBatch(schema, cols, n)
    return cpu_engine.encode_batch(batch, "Example"), schema


class TestMutationFuzz:
    def test_byte_flips_never_crash(self):
        img, schema = _image()
        rng = np.random.default_rng(42)
        base = np.frombuffer(img, np.uint8).copy()
        for trial in range(300):
            data = base.copy()
            for _ in range(int(rng.integers(1, 4))):
                data[rng.integers(0, len(data))] ^= int(rng.integers(1, 256))
            try:
                cpu_engine.decode_buffer(data, schema, "Example",
                                         verify_crc=True)
            except (RuntimeError, ValueError):
                pass  # rejected cleanly — the only acceptable failure mode

    def test_truncations_never_crash(self):
        img, schema = _image(seed=1)
        base = np.frombuffer(img, np.uint8)
        rng = np.random.default_rng(7)
        for trial in range(150):
            cut = int(rng.integers(0, len(base)))
            try:
                cpu_engine.decode_buffer(base[:cut].copy(), schema, "Example")
            except (RuntimeError, ValueError):
                pass

    def test_random_garbage_never_crashes(self):
        schema = stf.StructType([stf.StructField("x", stf.LongType(), True)])
        rng = np.random.default_rng(3)
        for trial in range(100):
            blob = rng.bytes(int(rng.integers(1, 5000)))
            try:
                cpu_engine.decode_buffer(np.frombuffer(blob, np.uint8), schema,
                                         "Example")
            except (RuntimeError, ValueError):
                pass

    def test_fused_and_plain_agree_on_mutations(self):
        img, schema = _image(seed=2)
        base = np.frombuffer(img, np.uint8).copy()
        blob = schema_blob(schema)
        rng = np.random.default_rng(9)
        for trial in range(150):
            data = base.copy()
            for _ in range(int(rng.integers(1, 3))):
                data[rng.integers(0, len(data))] ^= int(rng.integers(1, 256))
            try:
                off, lens = _native.scan_frames(data, False)
            except RuntimeError:
                continue
            plain = _native.scan_stats_debug(data, off, lens,
                                             _native.FMT_EXAMPLE, blob, False)
            fused = _native.scan_stats_debug(data, off, lens,
                                             _native.FMT_EXAMPLE, blob, True)
            # verdicts must agree at record level; on records BOTH accept,
            # per-field verdicts, clean-field stats and CRCs must be
            # identical (exact error codes / partial stats of rejected
            # records may differ — both forms still reject them)
            np.testing.assert_array_equal(plain["rc"] == 0, fused["rc"] == 0)
            ok = plain["rc"] == 0
            np.testing.assert_array_equal(plain["crc"][ok], fused["crc"][ok])
            pe = plain["stats"][:, :, 5] >> 32  # packed per-field err
            fe = fused["stats"][:, :, 5] >> 32
            np.testing.assert_array_equal((pe == 0)[ok], (fe == 0)[ok])
            clean = (pe == 0) & ok[:, None]
            np.testing.assert_array_equal(plain["stats"][clean],
                                          fused["stats"][clean])

    def test_mutated_inference_never_crashes(self):
        img, _ = _image(seed=4)
        base = np.frombuffer(img, np.uint8).copy()
        rng = np.random.default_rng(11)
        from spark_tfrecord_amd.infer import infer_codes_from_buffer
        for trial in range(100):
            data = base.copy()
            data[rng.integers(0, len(data))] ^= int(rng.integers(1, 256))
            try:
                off, lens = _native.scan_frames(data, False)
                infer_codes_from_buffer(data, off, lens, "Example")
            except (RuntimeError, ValueError):
                pass
