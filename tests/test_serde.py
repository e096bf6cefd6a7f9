"""Tier-1 serde tests, mirroring the reference's pure unit suites
(TFRecordSerializerTest.scala, TFRecordDeserializerTest.scala): every
scalar/array/nested type to feature kind + value, null handling, unsupported
types, kind mismatches, and the cross-row state-leak regression."""

import decimal

import numpy as np
import pytest

import spark_tfrecord_amd as stf
from spark_tfrecord_amd import _native
from spark_tfrecord_amd.columnar import (
    RecordBatch,
    column_from_values,
    column_to_pylist,
)
from spark_tfrecord_amd.engine import cpu as cpu_engine
from spark_tfrecord_amd.schema import (
    KIND_BYTES,
    KIND_FLOAT,
    KIND_INT64,
    merge_types,
    wire_kind_of,
)


def roundtrip(values, dtype, nullable=True, name="f", record_type="Example"):
    schema = stf.StructType([stf.StructField(name, dtype, nullable)])
    col = column_from_values(values, dtype, nullable, name)
    batch = RecordBatch(schema, [col], len(values))
    img = cpu_engine.encode_batch(batch, record_type)
    out = cpu_engine.decode_buffer(np.frombuffer(img, np.uint8), schema, record_type)
    return column_to_pylist(out.columns[0], dtype, nullable, name)


def parse_first(values, dtype, protos, nullable=True, record_type="Example"):
    schema = stf.StructType([stf.StructField("f", dtype, nullable)])
    col = column_from_values(values, dtype, nullable, "f")
    img = cpu_engine.encode_batch(RecordBatch(schema, [col], len(values)), record_type)
    off, ln = _native.scan_frames(np.frombuffer(img, np.uint8), True)
    cls = protos.SequenceExample if record_type == "SequenceExample" else protos.Example
    msgs = []
    for o, l in zip(off, ln):
        m = cls()
        m.ParseFromString(img[o:o + l])
        msgs.append(m)
    return msgs


class TestSerializerKinds:
    """Serializer maps every supported type to the right Feature kind
    (TFRecordSerializerTest.scala:71-141)."""

    def test_integer_to_int64list(self, tf_example_protos):
        (m,) = parse_first([7], stf.IntegerType(), tf_example_protos)
        assert list(m.features.feature["f"].int64_list.value) == [7]

    def test_long_to_int64list(self, tf_example_protos):
        (m,) = parse_first([2**40], stf.LongType(), tf_example_protos)
        assert list(m.features.feature["f"].int64_list.value) == [2**40]

    def test_negative_int64(self, tf_example_protos):
        (m,) = parse_first([-42], stf.LongType(), tf_example_protos)
        assert list(m.features.feature["f"].int64_list.value) == [-42]

    def test_float_to_floatlist(self, tf_example_protos):
        (m,) = parse_first([1.25], stf.FloatType(), tf_example_protos)
        assert m.features.feature["f"].float_list.value[0] == 1.25

    def test_double_downcast_to_float32(self, tf_example_protos):
        # reference: TFRecordSerializer.scala:86 — Double -> float32
        v = 0.1234567890123456789
        (m,) = parse_first([v], stf.DoubleType(), tf_example_protos)
        got = m.features.feature["f"].float_list.value[0]
        assert got == pytest.approx(v, abs=1e-7)
        assert got != v  # lossy by design

    def test_decimal_to_float32(self, tf_example_protos):
        (m,) = parse_first([decimal.Decimal("2.5")], stf.DecimalType(),
                           tf_example_protos)
        assert m.features.feature["f"].float_list.value[0] == 2.5

    def test_string_to_bytes(self, tf_example_protos):
        (m,) = parse_first(["héllo"], stf.StringType(), tf_example_protos)
        assert m.features.feature["f"].bytes_list.value[0] == "héllo".encode("utf-8")

    def test_binary_to_bytes(self, tf_example_protos):
        (m,) = parse_first([b"\x00\xff"], stf.BinaryType(), tf_example_protos)
        assert m.features.feature["f"].bytes_list.value[0] == b"\x00\xff"

    def test_long_array(self, tf_example_protos):
        (m,) = parse_first([[1, 2, 3]], stf.ArrayType(stf.LongType()),
                           tf_example_protos)
        assert list(m.features.feature["f"].int64_list.value) == [1, 2, 3]

    def test_string_array(self, tf_example_protos):
        (m,) = parse_first([["a", "b"]], stf.ArrayType(stf.StringType()),
                           tf_example_protos)
        assert list(m.features.feature["f"].bytes_list.value) == [b"a", b"b"]

    def test_empty_array(self, tf_example_protos):
        (m,) = parse_first([[]], stf.ArrayType(stf.LongType()), tf_example_protos)
        assert "f" in m.features.feature
        assert list(m.features.feature["f"].int64_list.value) == []

    def test_nested_array_to_featurelist(self, tf_example_protos):
        (m,) = parse_first([[[1.0, 2.0], [3.0]]],
                           stf.ArrayType(stf.ArrayType(stf.FloatType())),
                           tf_example_protos, record_type="SequenceExample")
        fl = m.feature_lists.feature_list["f"].feature
        assert len(fl) == 2
        assert list(fl[0].float_list.value) == [1.0, 2.0]
        assert list(fl[1].float_list.value) == [3.0]

    def test_sequence_context_features(self, tf_example_protos):
        schema = stf.StructType([
            stf.StructField("ctx", stf.LongType(), True),
            stf.StructField("seq", stf.ArrayType(stf.ArrayType(stf.LongType())), True),
        ])
        cols = [column_from_values([5], stf.LongType(), True, "ctx"),
                column_from_values([[[1], [2, 3]]], schema[1].dataType, True, "seq")]
        img = cpu_engine.encode_batch(RecordBatch(schema, cols, 1), "SequenceExample")
        off, ln = _native.scan_frames(np.frombuffer(img, np.uint8), True)
        m = tf_example_protos.SequenceExample()
        m.ParseFromString(img[off[0]:off[0] + ln[0]])
        assert m.context.feature["ctx"].int64_list.value[0] == 5
        assert len(m.feature_lists.feature_list["seq"].feature) == 2


class TestSerializerNulls:
    """Null semantics (TFRecordSerializerTest.scala:229-288)."""

    def test_nullable_null_omits_feature(self, tf_example_protos):
        m1, m2 = parse_first([None, 3], stf.LongType(), tf_example_protos)
        assert "f" not in m1.features.feature
        assert m2.features.feature["f"].int64_list.value[0] == 3

    def test_non_nullable_null_raises(self):
        with pytest.raises(ValueError, match="non-nullable"):
            column_from_values([None], stf.LongType(), False, "f")

    def test_unsupported_type_rejected_at_construction(self):
        with pytest.raises(TypeError):
            column_from_values([[[[1]]]],
                               stf.ArrayType(stf.ArrayType(stf.ArrayType(stf.LongType()))),
                               True, "f")
        with pytest.raises(TypeError):
            column_from_values([{}], stf.StructType([]), True, "f")


class TestDeserializer:
    """Mirror-image decode tests (TFRecordDeserializerTest.scala:61-162)."""

    def test_scalar_types_roundtrip(self):
        assert roundtrip([3, None], stf.LongType()) == [3, None]
        assert roundtrip([3], stf.IntegerType()) == [3]
        assert roundtrip([1.5], stf.FloatType()) == [1.5]
        assert roundtrip(["x"], stf.StringType()) == ["x"]
        assert roundtrip([b"\x01"], stf.BinaryType()) == [b"\x01"]

    def test_double_roundtrip_is_float32(self):
        v = 0.123456789
        (got,) = roundtrip([v], stf.DoubleType())
        assert got == pytest.approx(v, abs=1e-7)
        assert got == float(np.float32(v))

    def test_decimal_roundtrip(self):
        (got,) = roundtrip([decimal.Decimal("1.5")], stf.DecimalType())
        assert isinstance(got, decimal.Decimal)
        assert float(got) == 1.5

    def test_arrays_roundtrip(self):
        assert roundtrip([[1, 2], [], [3]], stf.ArrayType(stf.LongType())) == \
            [[1, 2], [], [3]]
        assert roundtrip([["a", "bb"]], stf.ArrayType(stf.StringType())) == \
            [["a", "bb"]]

    def test_nested_roundtrip(self):
        v = [[[1.0], [2.0, 3.0]], [[4.0]]]
        assert roundtrip(v, stf.ArrayType(stf.ArrayType(stf.FloatType())),
                         record_type="SequenceExample") == v

    def test_integer_downcasts_int64(self):
        schema = stf.StructType([stf.StructField("f", stf.LongType(), True)])
        col = column_from_values([5], stf.LongType(), True, "f")
        img = cpu_engine.encode_batch(RecordBatch(schema, [col], 1), "Example")
        int_schema = stf.StructType([stf.StructField("f", stf.IntegerType(), True)])
        out = cpu_engine.decode_buffer(np.frombuffer(img, np.uint8), int_schema,
                                       "Example")
        assert column_to_pylist(out.columns[0], stf.IntegerType(), True, "f") == [5]

    def test_kind_mismatch_raises(self):
        schema = stf.StructType([stf.StructField("f", stf.LongType(), True)])
        col = column_from_values([5], stf.LongType(), True, "f")
        img = cpu_engine.encode_batch(RecordBatch(schema, [col], 1), "Example")
        str_schema = stf.StructType([stf.StructField("f", stf.StringType(), True)])
        with pytest.raises(RuntimeError, match="kind"):
            cpu_engine.decode_buffer(np.frombuffer(img, np.uint8), str_schema,
                                     "Example")

    def test_non_nullable_missing_raises(self):
        schema = stf.StructType([stf.StructField("f", stf.LongType(), True),
                                 stf.StructField("g", stf.LongType(), True)])
        cols = [column_from_values([1], stf.LongType(), True, "f"),
                column_from_values([None], stf.LongType(), True, "g")]
        img = cpu_engine.encode_batch(RecordBatch(schema, cols, 1), "Example")
        out_schema = stf.StructType([stf.StructField("g", stf.LongType(), False)])
        out = cpu_engine.decode_buffer(np.frombuffer(img, np.uint8), out_schema,
                                       "Example")
        with pytest.raises(ValueError, match="required"):
            column_to_pylist(out.columns[0], stf.LongType(), False, "g")

    def test_nullable_missing_is_null(self):
        schema = stf.StructType([stf.StructField("f", stf.LongType(), True)])
        col = column_from_values([1], stf.LongType(), True, "f")
        img = cpu_engine.encode_batch(RecordBatch(schema, [col], 1), "Example")
        out_schema = stf.StructType([stf.StructField("missing", stf.LongType(), True)])
        out = cpu_engine.decode_buffer(np.frombuffer(img, np.uint8), out_schema,
                                       "Example")
        assert column_to_pylist(out.columns[0], stf.LongType(), True, "missing") == \
            [None]

    def test_no_state_leak_between_rows(self):
        """Rows must not inherit features from previous rows
        (regression mirror of TFRecordDeserializerTest.scala:313-346)."""
        vals = [[1, 2, 3], None, [9]]
        got = roundtrip(vals, stf.ArrayType(stf.LongType()))
        assert got == [[1, 2, 3], None, [9]]

    def test_unknown_features_ignored(self, tf_example_protos):
        e = tf_example_protos.Example()
        e.features.feature["known"].int64_list.value.append(1)
        e.features.feature["unknown"].float_list.value.append(9.0)
        payload = e.SerializeToString()
        arr = np.frombuffer(payload, np.uint8)
        schema = stf.StructType([stf.StructField("known", stf.LongType(), True)])
        from spark_tfrecord_amd.columnar import schema_blob
        out = _native.decode_records(arr, np.array([0], np.int64),
                                     np.array([len(payload)], np.int64),
                                     schema_blob(schema), _native.FMT_EXAMPLE)
        assert list(out[0]["values"]) == [1]


class TestLattice:
    """merge_types mirrors findTightestCommonType
    (TensorFlowInferSchema.scala:213-228)."""

    def test_numeric_promotion(self):
        assert merge_types(stf.LongType(), stf.FloatType()) == stf.FloatType()
        assert merge_types(stf.FloatType(), stf.StringType()) == stf.StringType()
        assert merge_types(stf.LongType(), stf.StringType()) == stf.StringType()

    def test_scalar_array_promotion(self):
        assert merge_types(stf.LongType(), stf.ArrayType(stf.LongType())) == \
            stf.ArrayType(stf.LongType())
        assert merge_types(stf.ArrayType(stf.LongType()),
                           stf.ArrayType(stf.FloatType())) == \
            stf.ArrayType(stf.FloatType())

    def test_null_merges(self):
        assert merge_types(stf.NullType(), stf.LongType()) == stf.LongType()
        assert merge_types(None, None) is None

    def test_kinds(self):
        assert wire_kind_of(stf.LongType()) == KIND_INT64
        assert wire_kind_of(stf.DoubleType()) == KIND_FLOAT
        assert wire_kind_of(stf.ArrayType(stf.StringType())) == KIND_BYTES


class TestArrowNestedFastPath:
    def test_nested_list_fast_path_matches_fallback(self):
        import numpy as np
        import pyarrow as pa

        from spark_tfrecord_amd.arrow_interop import arrow_to_wire
        from spark_tfrecord_amd.columnar import column_from_values

        rng = np.random.default_rng(0)
        rows = [[[float(v) for v in rng.random(int(rng.integers(0, 4)))]
                 for _ in range(int(rng.integers(0, 3)))] for _ in range(200)]
        dt = stf.ArrayType(stf.ArrayType(stf.FloatType()))
        arr = pa.array(rows, type=pa.large_list(pa.large_list(pa.float32())))
        fast = arrow_to_wire(arr, dt, True, "rag")
        slow = column_from_values(rows, dt, True, "rag")
        np.testing.assert_array_equal(fast.presence, slow.presence)
        np.testing.assert_array_equal(fast.row_off, slow.row_off)
        np.testing.assert_array_equal(np.asarray(fast.list_off),
                                      np.asarray(slow.list_off))
        np.testing.assert_array_equal(np.asarray(fast.sub_off),
                                      np.asarray(slow.sub_off))
        np.testing.assert_array_equal(np.asarray(fast.values),
                                      np.asarray(slow.values))

    def test_nested_with_nulls_uses_fallback(self):
        import numpy as np
        import pyarrow as pa

        from spark_tfrecord_amd.arrow_interop import arrow_to_wire

        rows = [[[1.0]], None, [[2.0, 3.0], []]]
        dt = stf.ArrayType(stf.ArrayType(stf.FloatType()))
        arr = pa.array(rows, type=pa.large_list(pa.large_list(pa.float32())))
        col = arrow_to_wire(arr, dt, True, "rag")
        np.testing.assert_array_equal(col.presence, [1, 0, 1])
        np.testing.assert_array_equal(np.asarray(col.row_off), [0, 1, 1, 3])


class TestFullTypeMatrix:
    """Every type combination the reference's README advertises (scalars,
    arrays, arrays-of-arrays of Integer/Long/Float/Double/Decimal/String/
    Binary) round-trips. Spark ML VectorType has no non-JVM analog; its
    ArrayType(DoubleType) representation is what this covers."""

    def test_all_advertised_combinations(self):
        from decimal import Decimal

        import numpy as np

        from spark_tfrecord_amd.columnar import RecordBatch, column_from_values
        from spark_tfrecord_amd.engine import cpu as cpu_engine

        cases = [
            (stf.ArrayType(stf.IntegerType()), [[1, 2], [], None, [5]]),
            (stf.ArrayType(stf.DoubleType()), [[1.5, 2.5], None, [0.25]]),
            (stf.ArrayType(stf.DecimalType()),
             [[Decimal("1.5")], [Decimal("2.25"), Decimal("0.5")]]),
            (stf.ArrayType(stf.BinaryType()), [[b"ab", b""], [b"c"]]),
            (stf.ArrayType(stf.ArrayType(stf.IntegerType())), [[[1], [2, 3]], []]),
            (stf.ArrayType(stf.ArrayType(stf.DoubleType())),
             [[[1.5]], [[2.5, 3.5], []]]),
            (stf.ArrayType(stf.ArrayType(stf.StringType())),
             [[["a", "bb"]], [["c"], []]]),
            (stf.ArrayType(stf.ArrayType(stf.BinaryType())),
             [[[b"x"]], [[b"yy", b"z"]]]),
        ]
        for dt, vals in cases:
            schema = stf.StructType([stf.StructField("c", dt, True)])
            col = column_from_values(vals, dt, True, "c")
            batch = RecordBatch(schema, [col], len(vals))
            rt = ("SequenceExample"
                  if isinstance(dt.elementType, stf.ArrayType) else "Example")
            img = cpu_engine.encode_batch(batch, rt)
            out = cpu_engine.decode_buffer(np.frombuffer(img, np.uint8), schema, rt)
            np.testing.assert_array_equal(np.asarray(col.values),
                                          np.asarray(out.columns[0].values))


class TestEdgeSemanticsParity:
    """Pin edge-case semantics to the reference's behavior."""

    def test_int64_to_integer_silent_truncation(self):
        """IntegerType reads of out-of-range Int64List values truncate like
        the reference's Scala .toInt (TFRecordDeserializer.scala:84-86) —
        numpy astype(int32) has identical wraparound semantics."""
        from spark_tfrecord_amd.columnar import column_from_values, column_to_pylist
        from spark_tfrecord_amd.engine import cpu as cpu_engine
        big = 2**31 + 5  # wraps to -2**31 + 5 in int32
        schema = stf.StructType([stf.StructField("f", stf.LongType(), True)])
        b = RecordBatch(schema, [column_from_values(
            [big], stf.LongType(), True, "f")], 1)
        raw = cpu_engine.encode_batch(b, "Example")
        int_schema = stf.StructType([stf.StructField("f", stf.IntegerType(), True)])
        out = cpu_engine.decode_buffer(
            np.frombuffer(raw, np.uint8), int_schema, "Example")
        assert column_to_pylist(out.columns[0], stf.IntegerType(), True, "f") \
            == [np.int64(big).astype(np.int32).item()]

    def test_megabyte_single_string(self, tmp_path):
        s = "x" * (1 << 20) + "end"
        out = str(tmp_path / "big")
        stf.write_tfrecord({"s": [s, "tiny"]}, out, engine="cpu")
        got = stf.read_tfrecord(out, engine="cpu").collect()
        assert sorted(r["s"] for r in got) == sorted([s, "tiny"])

    def test_unicode_feature_names(self, tmp_path):
        out = str(tmp_path / "uni")
        stf.write_tfrecord({"héllo_名前": np.arange(10, dtype=np.int64)},
                           out, engine="cpu")
        df = stf.read_tfrecord(out, engine="cpu")
        assert df.columns == ["héllo_名前"]
        assert df.count() == 10

    def test_empty_list_vs_missing_distinction(self, tmp_path):
        """An EMPTY Int64List feature reads as an empty array; a MISSING
        feature reads as null (TFRecordDeserializer.scala nullable rules +
        parseInt64List length-0 semantics)."""
        import pyarrow as pa
        out = str(tmp_path / "el")
        t = pa.table({"a": pa.array([[1, 2], [], None],
                                    type=pa.large_list(pa.int64()))})
        schema = stf.StructType([
            stf.StructField("a", stf.ArrayType(stf.LongType()), True)])
        stf.write_tfrecord(t, out, schema=schema, engine="cpu")
        got = stf.read_tfrecord(out, schema=schema, engine="cpu") \
            .to_arrow_table().column("a").to_pylist()
        # written [] keeps its presence as an empty Int64List feature; a
        # written None is OMITTED from the record (reference serializer
        # null rule) and reads back as null
        assert got[0] == [1, 2]
        # [] was written as an empty feature -> empty array on read
        assert got[1] == []
        # None was omitted -> null on read
        assert got[2] is None


def _raw_example_with_dup_keys():
    """Hand-built Example whose Features map repeats the key 'x' — valid
    protobuf; map semantics are LAST-entry-wins (what protobuf-java gives
    the reference). Returns a framed single-record file image."""
    import struct

    def varint(v):
        out = b""
        while v >= 0x80:
            out += bytes([v & 0x7F | 0x80])
            v >>= 7
        return out + bytes([v])

    def feature_int64(vals):
        packed = b"".join(varint(x) for x in vals)
        body = bytes([0x0a]) + varint(len(packed)) + packed
        return bytes([0x1a]) + varint(len(body)) + body

    def map_entry(key, feat):
        e = (bytes([0x0a]) + varint(len(key)) + key.encode() +
             bytes([0x12]) + varint(len(feat)) + feat)
        return bytes([0x0a]) + varint(len(e)) + e

    entries = (map_entry("x", feature_int64([1, 2])) +
               map_entry("x", feature_int64([7])))
    example = bytes([0x0a]) + varint(len(entries)) + entries

    def mask(c):
        return ((c >> 15) | (c << 17)) + 0xa282ead8 & 0xFFFFFFFF

    ln = struct.pack("<Q", len(example))
    return (ln + struct.pack("<I", mask(_native.crc32c(ln))) + example +
            struct.pack("<I", mask(_native.crc32c(example))))


class TestDuplicateMapKeys:
    def test_last_entry_wins(self):
        """Duplicate Features-map keys must decode as the LAST entry only —
        the scan previously accumulated counts across duplicates while the
        extract read only the final body, leaving garbage slots."""
        img = _raw_example_with_dup_keys()
        schema = stf.StructType([
            stf.StructField("x", stf.ArrayType(stf.LongType()), True)])
        out = cpu_engine.decode_buffer(np.frombuffer(bytearray(img), np.uint8),
                                       schema, "Example")
        c = out.columns[0]
        assert list(np.asarray(c.row_off)) == [0, 1]
        assert list(np.asarray(c.values)) == [7]
