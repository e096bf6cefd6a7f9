"""Schema-inference tests, mirroring InferSchemaSuite.scala: type promotion
across rows, SequenceExample 2-D inference, empty lists -> NullType, and the
first-non-empty-file rule."""

import numpy as np
import pytest

import spark_tfrecord_amd as stf
from spark_tfrecord_amd import _native
from spark_tfrecord_amd.infer import (
    infer_codes_from_buffer,
    merge_code_maps,
    schema_from_codes,
)


def build_file(protos, examples, path=None):
    """Serialize protobuf-built Examples into an in-memory TFRecord image."""
    import struct

    def crc32c_ref(data):
        crc = 0xFFFFFFFF
        for b in data:
            crc ^= b
            for _ in range(8):
                crc = (crc >> 1) ^ (0x82F63B78 if crc & 1 else 0)
        return crc ^ 0xFFFFFFFF

    def mask(c):
        return (((c >> 15) | (c << 17)) + 0xA282EAD8) & 0xFFFFFFFF

    out = b""
    for e in examples:
        p = e.SerializeToString()
        h = struct.pack("<Q", len(p))
        out += h + struct.pack("<I", mask(crc32c_ref(h))) + p + \
            struct.pack("<I", mask(crc32c_ref(p)))
    if path is not None:
        path.write_bytes(out)
    return np.frombuffer(out, np.uint8)


def infer(protos, examples, record_type="Example"):
    data = build_file(protos, examples)
    off, ln = _native.scan_frames(data, True)
    codes = infer_codes_from_buffer(data, off, ln, record_type)
    return schema_from_codes(codes)


def ex(protos, **feats):
    e = protos.Example()
    for name, vals in feats.items():
        if not isinstance(vals, (list, tuple)):
            vals = [vals]
        if all(isinstance(v, int) for v in vals):
            e.features.feature[name].int64_list.value.extend(vals)
        elif all(isinstance(v, float) for v in vals):
            e.features.feature[name].float_list.value.extend(vals)
        else:
            e.features.feature[name].bytes_list.value.extend(
                v.encode() if isinstance(v, str) else v for v in vals)
    return e


class TestExampleInference:
    def test_scalar_types(self, tf_example_protos):
        s = infer(tf_example_protos, [ex(tf_example_protos, a=1, b=2.0, c="x")])
        assert s["a"].dataType == stf.LongType()
        assert s["b"].dataType == stf.FloatType()
        assert s["c"].dataType == stf.StringType()

    def test_multi_element_infers_array(self, tf_example_protos):
        s = infer(tf_example_protos, [ex(tf_example_protos, a=[1, 2])])
        assert s["a"].dataType == stf.ArrayType(stf.LongType())

    def test_mixed_long_float_promotes_to_float(self, tf_example_protos):
        # InferSchemaSuite.scala:39-79 MixedTypeList
        s = infer(tf_example_protos,
                  [ex(tf_example_protos, a=[1, 2]),
                   ex(tf_example_protos, a=[0.5, 0.25])])
        assert s["a"].dataType == stf.ArrayType(stf.FloatType())

    def test_scalar_float_with_long_array_takes_lattice_max(self, tf_example_protos):
        # Lattice max, not element promotion: Float(2) vs Arr[Long](4) -> Arr[Long]
        # (findTightestCommonType picks the higher precedence, TensorFlowInferSchema.scala:213-228)
        s = infer(tf_example_protos,
                  [ex(tf_example_protos, a=[1, 2]), ex(tf_example_protos, a=[0.5])])
        assert s["a"].dataType == stf.ArrayType(stf.LongType())

    def test_mixed_scalar_array_promotes_to_array(self, tf_example_protos):
        s = infer(tf_example_protos,
                  [ex(tf_example_protos, a=1), ex(tf_example_protos, a=[1, 2])])
        assert s["a"].dataType == stf.ArrayType(stf.LongType())

    def test_long_string_promotes_to_string(self, tf_example_protos):
        s = infer(tf_example_protos,
                  [ex(tf_example_protos, a=1), ex(tf_example_protos, a="x")])
        assert s["a"].dataType == stf.StringType()

    def test_empty_feature_infers_null(self, tf_example_protos):
        # empty list => NullType (InferSchemaSuite.scala:142-155)
        e = tf_example_protos.Example()
        e.features.feature["a"].int64_list.SetInParent()
        s = infer(tf_example_protos, [e])
        assert s["a"].dataType == stf.NullType()

    def test_null_then_value_merges(self, tf_example_protos):
        e = tf_example_protos.Example()
        e.features.feature["a"].int64_list.SetInParent()
        s = infer(tf_example_protos, [e, ex(tf_example_protos, a=3)])
        assert s["a"].dataType == stf.LongType()

    def test_missing_in_some_rows(self, tf_example_protos):
        s = infer(tf_example_protos,
                  [ex(tf_example_protos, a=1), ex(tf_example_protos, b=2.0)])
        assert s["a"].dataType == stf.LongType()
        assert s["b"].dataType == stf.FloatType()


class TestSequenceInference:
    def seq(self, protos, ctx=None, lists=None):
        se = protos.SequenceExample()
        if ctx:
            for k, v in ctx.items():
                se.context.feature[k].int64_list.value.append(v)
        if lists:
            for k, sublists in lists.items():
                fl = se.feature_lists.feature_list[k]
                for sub in sublists:
                    f = fl.feature.add()
                    if all(isinstance(x, float) for x in sub):
                        f.float_list.value.extend(sub)
                    elif all(isinstance(x, int) for x in sub):
                        f.int64_list.value.extend(sub)
                    else:
                        f.bytes_list.value.extend(
                            x.encode() if isinstance(x, str) else x for x in sub)
        return se

    def test_feature_list_infers_2d(self, tf_example_protos):
        se = self.seq(tf_example_protos, lists={"fl": [[1.0, 2.0], [3.0]]})
        data = build_file(tf_example_protos, [se])
        off, ln = _native.scan_frames(data, True)
        s = schema_from_codes(infer_codes_from_buffer(data, off, ln,
                                                      "SequenceExample"))
        assert s["fl"].dataType == stf.ArrayType(stf.ArrayType(stf.FloatType()))

    def test_mixed_kinds_promote_to_string_2d(self, tf_example_protos):
        # InferSchemaSuite.scala:81-132 Mixed => Arr[Arr[String]]
        se1 = self.seq(tf_example_protos, lists={"fl": [[1, 2]]})
        se2 = self.seq(tf_example_protos, lists={"fl": [["a"]]})
        data = build_file(tf_example_protos, [se1, se2])
        off, ln = _native.scan_frames(data, True)
        s = schema_from_codes(infer_codes_from_buffer(data, off, ln,
                                                      "SequenceExample"))
        assert s["fl"].dataType == stf.ArrayType(stf.ArrayType(stf.StringType()))

    def test_context_and_lists_together(self, tf_example_protos):
        se = self.seq(tf_example_protos, ctx={"c": 1}, lists={"fl": [[1]]})
        data = build_file(tf_example_protos, [se])
        off, ln = _native.scan_frames(data, True)
        s = schema_from_codes(infer_codes_from_buffer(data, off, ln,
                                                      "SequenceExample"))
        assert s["c"].dataType == stf.LongType()
        assert s["fl"].dataType == stf.ArrayType(stf.ArrayType(stf.LongType()))


class TestInferencePlumbing:
    def test_merge_code_maps(self):
        merged = merge_code_maps([{"a": 1, "b": 2}, {"a": 4, "c": 0}])
        assert merged == {"a": 4, "b": 2, "c": 0}

    def test_first_non_empty_file_rule(self, tf_example_protos, tmp_sandbox):
        # DefaultSource.scala:36-38: schema from the first non-empty file only
        from spark_tfrecord_amd.io.reader import infer_schema_of_paths

        p0 = tmp_sandbox / "0.tfrecord"
        p0.write_bytes(b"")
        p1 = tmp_sandbox / "1.tfrecord"
        build_file(tf_example_protos, [ex(tf_example_protos, a=1)], p1)
        p2 = tmp_sandbox / "2.tfrecord"
        build_file(tf_example_protos, [ex(tf_example_protos, a="str")], p2)
        s = infer_schema_of_paths([str(p0), str(p1), str(p2)], "Example")
        # p2's string never seen: schema comes from p1 alone
        assert s["a"].dataType == stf.LongType()

    def test_read_applies_inferred_schema(self, tf_example_protos, tmp_sandbox):
        p = tmp_sandbox / "f.tfrecord"
        build_file(tf_example_protos,
                   [ex(tf_example_protos, a=1.0, b=[1.0, 2.0]),
                    ex(tf_example_protos, a=2.5)], p)
        df = stf.read_tfrecord(str(p))
        assert df.schema["a"].dataType == stf.FloatType()
        assert df.schema["b"].dataType == stf.ArrayType(stf.FloatType())
        rows = df.collect()
        assert rows[0]["a"] == 1.0 and rows[1]["b"] is None

    def test_promoted_schema_is_strict_on_read(self, tf_example_protos, tmp_sandbox):
        """Inference can promote Long+Float rows to Float, but the decoder is
        kind-strict like the reference's accessors
        (TFRecordDeserializer.scala:177-199 require(kind == ...)): reading the
        int64 rows under the promoted Float schema raises."""
        p = tmp_sandbox / "g.tfrecord"
        build_file(tf_example_protos,
                   [ex(tf_example_protos, a=[1, 2]),
                    ex(tf_example_protos, a=[0.5, 0.25])], p)
        with pytest.raises(RuntimeError, match="kind"):
            stf.read_tfrecord(str(p)).collect()
