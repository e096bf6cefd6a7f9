"""Tier-1 wire-format tests: framing + masked CRC32C, against an independent
pure-Python implementation and google.protobuf (cf. reference test strategy,
SURVEY.md §4 — serde testable without any runtime)."""

import struct

import numpy as np
import pytest

from spark_tfrecord_amd import _native

# -- independent pure-python CRC32C (bit-by-bit, different algorithm family
#    than the native slicing-by-8) --------------------------------------------


def crc32c_ref(data: bytes) -> int:
    crc = 0xFFFFFFFF
    for b in data:
        crc ^= b
        for _ in range(8):
            crc = (crc >> 1) ^ (0x82F63B78 if crc & 1 else 0)
    return crc ^ 0xFFFFFFFF


def mask_ref(crc: int) -> int:
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


def frame_ref(payload: bytes) -> bytes:
    """Reference TFRecord framing (SURVEY.md §1 on-disk format)."""
    header = struct.pack("<Q", len(payload))
    return (header + struct.pack("<I", mask_ref(crc32c_ref(header))) + payload +
            struct.pack("<I", mask_ref(crc32c_ref(payload))))


class TestCrc32c:
    def test_known_vectors(self):
        # RFC 3720 / common CRC32C test vectors
        assert _native.crc32c(b"123456789") == 0xE3069283
        assert _native.crc32c(b"") == 0
        assert _native.crc32c(b"\x00" * 32) == 0x8A9136AA
        assert _native.crc32c(b"\xff" * 32) == 0x62A8AB43

    def test_against_independent_impl(self):
        rng = np.random.default_rng(0)
        for n in [1, 3, 7, 8, 9, 63, 64, 65, 1000]:
            data = rng.integers(0, 256, n, dtype=np.uint8).tobytes()
            assert _native.crc32c(data) == crc32c_ref(data)

    def test_masking(self):
        for data in [b"abc", b"123456789", b"\x00" * 16]:
            assert _native.masked_crc32c(data) == mask_ref(crc32c_ref(data))


class TestFraming:
    def test_scan_accepts_reference_frames(self):
        payloads = [b"hello", b"", b"x" * 1000]
        blob = b"".join(frame_ref(p) for p in payloads)
        off, ln = _native.scan_frames(np.frombuffer(blob, np.uint8), True)
        assert list(ln) == [len(p) for p in payloads]
        for o, l, p in zip(off, ln, payloads):
            assert blob[o:o + l] == p

    def test_frame_byte_arrays_matches_reference(self):
        payloads = [b"alpha", b"", b"bravo-charlie"]
        data = np.frombuffer(b"".join(payloads), np.uint8)
        elem_off = np.cumsum([0] + [len(p) for p in payloads]).astype(np.int64)
        framed = _native.frame_byte_arrays(data, elem_off)
        assert bytes(framed) == b"".join(frame_ref(p) for p in payloads)

    def test_bad_length_crc_detected(self):
        blob = bytearray(frame_ref(b"payload"))
        blob[9] ^= 0xFF
        with pytest.raises(RuntimeError, match="bad length CRC"):
            _native.scan_frames(np.frombuffer(bytes(blob), np.uint8), True)

    def test_bad_data_crc_detected(self):
        blob = bytearray(frame_ref(b"payload"))
        blob[14] ^= 0x01  # flip a payload byte
        with pytest.raises(RuntimeError, match="bad data CRC"):
            _native.scan_frames(np.frombuffer(bytes(blob), np.uint8), True)

    def test_truncated_detected(self):
        blob = frame_ref(b"payload")[:-2]
        with pytest.raises(RuntimeError, match="truncated"):
            _native.scan_frames(np.frombuffer(blob, np.uint8), True)

    def test_verify_off_skips_crc(self):
        blob = bytearray(frame_ref(b"payload"))
        blob[14] ^= 0x01
        off, ln = _native.scan_frames(np.frombuffer(bytes(blob), np.uint8), False)
        assert len(off) == 1


class TestProtoInterop:
    def test_emitted_record_parses_with_protobuf(self, tf_example_protos):
        import spark_tfrecord_amd as stf
        from spark_tfrecord_amd.columnar import RecordBatch, column_from_values
        from spark_tfrecord_amd.engine import cpu as cpu_engine

        schema = stf.StructType([
            stf.StructField("ids", stf.ArrayType(stf.LongType()), True),
            stf.StructField("w", stf.FloatType(), True),
            stf.StructField("tag", stf.StringType(), True),
        ])
        cols = [
            column_from_values([[1, -2, 3], [10]], schema[0].dataType, True, "ids"),
            column_from_values([0.5, 1.5], stf.FloatType(), True, "w"),
            column_from_values(["a", "bc"], stf.StringType(), True, "tag"),
        ]
        img = cpu_engine.encode_batch(RecordBatch(schema, cols, 2), "Example")
        off, ln = _native.scan_frames(np.frombuffer(img, np.uint8), True)
        e = tf_example_protos.Example()
        e.ParseFromString(img[off[0]:off[0] + ln[0]])
        assert list(e.features.feature["ids"].int64_list.value) == [1, -2, 3]
        assert e.features.feature["w"].float_list.value[0] == 0.5
        assert e.features.feature["tag"].bytes_list.value[0] == b"a"

    def test_protobuf_built_file_decodes(self, tf_example_protos, tmp_sandbox):
        import spark_tfrecord_amd as stf

        records = []
        for i in range(5):
            e = tf_example_protos.Example()
            e.features.feature["x"].int64_list.value.append(i)
            e.features.feature["s"].bytes_list.value.append(f"v{i}".encode())
            records.append(e.SerializeToString())
        path = tmp_sandbox / "pb.tfrecord"
        path.write_bytes(b"".join(frame_ref(r) for r in records))
        df = stf.read_tfrecord(str(path))
        rows = df.sort("x").collect()
        assert [r["x"] for r in rows] == list(range(5))
        assert rows[2]["s"] == "v2"


class TestCrcCombine:
    def test_combine_matches_concatenation(self):
        import numpy as np

        from spark_tfrecord_amd import _native

        rng = np.random.default_rng(0)
        for _ in range(60):
            a = rng.bytes(int(rng.integers(0, 3000)))
            b = rng.bytes(int(rng.integers(0, 3000)))
            want = _native.crc32c(np.frombuffer(a + b, np.uint8))
            got = _native.crc32c_combine(
                _native.crc32c(np.frombuffer(a, np.uint8)),
                _native.crc32c(np.frombuffer(b, np.uint8)), len(b))
            assert got == want


class TestCrcCombineFast:
    def test_field_form_matches_matrix_and_direct(self):
        """crc32c_combine_fast (register-only GF(2^32) multiply, the wave
        kernels' fold) must equal the matrix form and the direct CRC of the
        concatenation for arbitrary splits."""
        import numpy as np
        from spark_tfrecord_amd import _native

        rng = np.random.default_rng(11)
        for n in (0, 1, 2, 7, 8, 9, 63, 64, 65, 255, 4096, 100_001):
            blob = rng.integers(0, 256, max(n, 1)).astype(np.uint8)[:n].tobytes()
            for cut in {0, n // 3, n // 2, n - 1 if n else 0, n}:
                a, b = blob[:cut], blob[cut:]
                want = _native.crc32c(blob)
                ca, cb = _native.crc32c(a), _native.crc32c(b)
                assert _native.crc32c_combine(ca, cb, len(b)) == want
                assert _native.crc32c_combine_fast(ca, cb, len(b)) == want
