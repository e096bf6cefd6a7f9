"""ShardWriter: incremental append semantics, atomic publish, gzip streaming."""

import gzip
import os

import numpy as np
import pytest

import spark_tfrecord_amd as stf


class TestShardWriter:
    def test_incremental_chunks_roundtrip(self, tmp_sandbox):
        p = str(tmp_sandbox / "sw" / "part-00000.tfrecord")
        with stf.ShardWriter(p, engine="cpu") as w:
            for k in range(5):
                n = w.write({"x": np.arange(k * 10, k * 10 + 10, dtype=np.int64)})
            assert n == 50
            assert not os.path.exists(p)  # nothing visible until close
        df = stf.read_tfrecord(p, engine="cpu").sort("x")
        assert [r["x"] for r in df.collect()] == list(range(50))

    def test_gzip_streaming(self, tmp_sandbox):
        p = str(tmp_sandbox / "swz" / "part-00000.tfrecord.gz")
        with stf.ShardWriter(p, codec="gzip", engine="cpu") as w:
            for k in range(3):
                w.write({"x": np.arange(20, dtype=np.int64) + 100 * k})
        # one valid gzip member any reader can decode
        raw = gzip.decompress(open(p, "rb").read())
        assert len(raw) > 0
        assert stf.count_tfrecord(str(tmp_sandbox / "swz"), engine="cpu") == 60

    def test_abort_leaves_nothing(self, tmp_sandbox):
        p = str(tmp_sandbox / "swa" / "part-00000.tfrecord")
        with pytest.raises(ValueError):
            with stf.ShardWriter(p, engine="cpu") as w:
                w.write({"x": np.arange(5, dtype=np.int64)})
                raise ValueError("boom")
        assert not os.path.exists(p)
        assert not os.path.exists(p + ".inprogress")

    def test_schema_locked_after_first_chunk(self, tmp_sandbox):
        p = str(tmp_sandbox / "swl" / "part-00000.tfrecord")
        with stf.ShardWriter(p, engine="cpu") as w:
            w.write({"x": np.arange(3, dtype=np.int64)})
            with pytest.raises(KeyError):
                w.write({"y": np.arange(3, dtype=np.int64)})
            w.write({"x": np.arange(3, dtype=np.int64)})
        assert stf.read_tfrecord(p, engine="cpu").count() == 6


from spark_tfrecord_amd.io import paths as P  # noqa: E402


class TestStreamedGzipSegmentTable:
    """Streamed gzip shards carry the same FEXTRA 'TS' segment table as
    write_tfrecord's gzip (backpatched into a reserved header region on
    close), so reads inflate them on the device. Overflowing the reserved
    capacity degrades to a table-less (marker-scan) file, like foreign
    gzip."""

    def _write(self, path, chunks, cap=2048):
        from spark_tfrecord_amd.io.stream_writer import ShardWriter
        with ShardWriter(path, record_type="Example", codec="gzip",
                         engine="cpu", segment_table_capacity=cap) as w:
            for c in chunks:
                w.write(c)

    def test_table_present_and_exact(self, tmp_path):
        import gzip as _gzip
        from spark_tfrecord_amd import _native
        p = str(tmp_path / "part-00000.tfrecord.gz")
        self._write(p, [{"x": np.arange(50_000, dtype=np.int64)}
                        for _ in range(3)])
        meta = P.parse_gz_segments_file(p)
        assert meta is not None
        body_off, segs, _crc, isize = meta
        raw = open(p, "rb").read()
        dec = _gzip.decompress(raw)
        assert len(dec) % (1 << 32) == isize
        pos, upos = body_off, 0
        for c, u in segs:
            assert _native.host_inflate_segment(raw[pos:pos + c], u) == \
                dec[upos:upos + u]
            pos += c
            upos += u
        assert pos == len(raw) - 8

    def test_capacity_overflow_falls_back_to_markers(self, tmp_path):
        p = str(tmp_path / "part-00000.tfrecord.gz")
        # tiny capacity: the shard exceeds it, table must be dropped
        self._write(p, [{"x": np.arange(80_000, dtype=np.int64)}], cap=2)
        assert P.parse_gz_segments_file(p) is None  # no (valid) TS table
        df = stf.read_tfrecord(str(tmp_path), engine="cpu")  # marker scan
        assert df.count() == 80_000

    def test_empty_shard(self, tmp_path):
        import gzip as _gzip
        from spark_tfrecord_amd.io.stream_writer import ShardWriter
        p = str(tmp_path / "part-00000.tfrecord.gz")
        w = ShardWriter(p, record_type="Example", codec="gzip", engine="cpu")
        w.close()
        assert _gzip.decompress(open(p, "rb").read()) == b""
        assert P.parse_gz_segments_file(p) is not None

    def test_boundary_aligned_chunks(self, tmp_path):
        # chunk sizes that land exactly on 32 KiB segment boundaries
        import gzip as _gzip
        p = str(tmp_path / "part-00000.tfrecord.gz")
        rows_32k = None
        from spark_tfrecord_amd.engine import cpu as cpu_engine
        from spark_tfrecord_amd.arrow_interop import table_to_batch, schema_from_arrow
        import pyarrow as pa
        # build a chunk whose encoded size is exactly one segment
        t = pa.table({"b": pa.array([b"z" * 100] * 273, type=pa.large_binary())})
        enc = cpu_engine.encode_batch(
            table_to_batch(t, schema_from_arrow(t.schema)), "Example")
        pad = P._GZ_SEGMENT - (len(enc) % P._GZ_SEGMENT)
        t2 = pa.table({"b": pa.array([b"y" * (pad - 21)], type=pa.large_binary())})
        from spark_tfrecord_amd.io.stream_writer import ShardWriter
        with ShardWriter(p, record_type="Example", codec="gzip",
                         engine="cpu") as w:
            w.write(t)
            w.write(t2)
        raw = open(p, "rb").read()
        dec = _gzip.decompress(raw)
        meta = P.parse_gz_segments_file(p)
        assert meta is not None
        assert sum(u for _, u in meta[1]) == len(dec)
