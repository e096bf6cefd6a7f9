"""Property-based round-trip tests (hypothesis): random schemas and values
through the CPU engine (the golden reference for the GPU kernels) must
round-trip exactly — same philosophy as the reference's epsilon-free integer
paths plus its lossy-float rules (TestingUtils.scala:30-121)."""

import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

import spark_tfrecord_amd as stf
from spark_tfrecord_amd.columnar import RecordBatch, column_from_values
from spark_tfrecord_amd.engine import cpu as cpu_engine

TYPES = [
    ("long", stf.LongType(), st.integers(min_value=-(2**63), max_value=2**63 - 1)),
    ("float", stf.FloatType(),
     st.floats(width=32, allow_nan=False, allow_infinity=False)),
    ("string", stf.StringType(), st.text(max_size=20)),
    ("binary", stf.BinaryType(), st.binary(max_size=20)),
]


def _column_strategy():
    def build(i_and_rows):
        i, rows = i_and_rows
        name, dtype, vals = TYPES[i % len(TYPES)]
        as_array = i % 2 == 1
        if as_array:
            elem = st.lists(vals, max_size=4)
            dtype = stf.ArrayType(dtype)
        else:
            elem = vals
        return st.tuples(
            st.just((f"c{i}_{name}", dtype)),
            st.lists(st.one_of(st.none(), elem), min_size=rows, max_size=rows))

    return build


@st.composite
def batches(draw):
    rows = draw(st.integers(min_value=0, max_value=30))
    ncols = draw(st.integers(min_value=1, max_value=4))
    fields = []
    cols = []
    for i in range(ncols):
        (name, dtype), values = draw(_column_strategy()((i, rows)))
        fields.append(stf.StructField(name, dtype, True))
        cols.append(column_from_values(values, dtype, True, name))
    return RecordBatch(stf.StructType(fields), cols, rows)


def assert_round_trip(batch, record_type):
    img = cpu_engine.encode_batch(batch, record_type)
    out = cpu_engine.decode_buffer(np.frombuffer(img, np.uint8), batch.schema,
                                   record_type)
    assert out.num_rows == batch.num_rows
    for ca, cb in zip(batch.columns, out.columns):
        np.testing.assert_array_equal(np.asarray(ca.presence), np.asarray(cb.presence))
        np.testing.assert_array_equal(np.asarray(ca.row_off), np.asarray(cb.row_off))
        np.testing.assert_array_equal(np.asarray(ca.values), np.asarray(cb.values))


class TestPropertyRoundTrip:
    @settings(max_examples=40, deadline=None)
    @given(batches())
    def test_example_roundtrip(self, batch):
        assert_round_trip(batch, "Example")

    @settings(max_examples=25, deadline=None)
    @given(st.lists(st.lists(st.lists(
        st.floats(width=32, allow_nan=False, allow_infinity=False),
        max_size=3), max_size=3), max_size=12))
    def test_sequence_ragged_roundtrip(self, rag):
        dtype = stf.ArrayType(stf.ArrayType(stf.FloatType()))
        schema = stf.StructType([stf.StructField("rag", dtype, True)])
        col = column_from_values(rag, dtype, True, "rag")
        batch = RecordBatch(schema, [col], len(rag))
        assert_round_trip(batch, "SequenceExample")

    @settings(max_examples=25, deadline=None)
    @given(st.lists(st.binary(max_size=64), max_size=12))
    def test_bytearray_roundtrip(self, payloads):
        import pyarrow as pa

        from spark_tfrecord_amd.arrow_interop import table_to_batch
        from spark_tfrecord_amd.infer import byte_array_schema

        table = pa.table({"byteArray": pa.array(payloads, type=pa.large_binary())})
        batch = table_to_batch(table, byte_array_schema())
        assert_round_trip(batch, "ByteArray")

    @settings(max_examples=25, deadline=None)
    @given(batches())
    def test_inference_agrees_with_decode(self, batch):
        """Inferred schema must itself decode the bytes it was inferred from."""
        from spark_tfrecord_amd import _native
        from spark_tfrecord_amd.infer import (infer_codes_from_buffer,
                                              schema_from_codes)

        img = cpu_engine.encode_batch(batch, "Example")
        data = np.frombuffer(img, np.uint8)
        if data.size == 0:
            return
        off, lens = _native.scan_frames(data, False)
        codes = infer_codes_from_buffer(data, off, lens, "Example")
        inferred = schema_from_codes(codes)
        out = cpu_engine.decode_buffer(data, inferred, "Example")
        assert out.num_rows == batch.num_rows
