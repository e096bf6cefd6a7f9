"""Gzip segment-table format (round 2): the engine writes standard gzip
whose FEXTRA field carries per-segment extents so readers (host pool and
the GPU inflater) decompress segments in parallel with exact offsets.
Interop contract: ANY gzip reader must decode the files unchanged."""

import gzip
import os
import zlib

import numpy as np
import pytest

import spark_tfrecord_amd as stf
from spark_tfrecord_amd.io import paths as P


def blob(n, seed=0, compressible=True):
    rng = np.random.default_rng(seed)
    if compressible:
        return bytes(rng.integers(65, 70, n).astype(np.uint8))
    return rng.bytes(n)  # random: zlib falls back to stored blocks


class TestSegmentTableFormat:
    def test_standard_gzip_reads_our_output(self):
        for n in (0, 1, 1000, P._GZ_SEGMENT, P._GZ_SEGMENT + 1,
                  3 * P._GZ_SEGMENT + 17):
            data = blob(n, seed=n % 7)
            gz = P.compress_bytes(data, "gzip")
            assert gzip.decompress(gz) == data
            assert zlib.decompress(gz, 16 + 15) == data

    def test_table_parse_roundtrip(self):
        data = blob(3 * P._GZ_SEGMENT + 999, seed=3)
        gz = P.compress_bytes(data, "gzip")
        meta = P.parse_gz_segments(gz)
        assert meta is not None
        body_off, segs, crc, isize = meta
        assert len(segs) == 4
        assert sum(u for _, u in segs) == len(data)
        assert isize == len(data) % (1 << 32)
        assert crc == (zlib.crc32(data) & 0xFFFFFFFF)
        # every segment is an independently inflatable raw-deflate stream
        pos = body_off
        out = b""
        for c, u in segs:
            d = zlib.decompressobj(-15)
            piece = d.decompress(gz[pos:pos + c]) + d.flush()
            assert len(piece) == u
            out += piece
            pos += c
        assert out == data

    def test_parse_file_variant(self, tmp_path):
        data = blob(2 * P._GZ_SEGMENT + 5, seed=9)
        p = str(tmp_path / "x.gz")
        with open(p, "wb") as f:
            f.write(P.compress_bytes(data, "gzip"))
        meta = P.parse_gz_segments_file(p)
        assert meta is not None and len(meta[1]) == 3
        assert P.parse_gz_segments_file(__file__) is None

    def test_host_parallel_inflate_uses_table(self):
        data = blob(5 * P._GZ_SEGMENT, seed=1, compressible=False)
        gz = P.compress_bytes(data, "gzip")
        assert P._gunzip_parallel(gz) == data

    def test_foreign_gzip_still_reads(self):
        data = blob(100_000, seed=4)
        assert P.decompress_file.__name__  # sanity
        foreign = gzip.compress(data, 6)
        assert P.parse_gz_segments(foreign) is None
        # decompress_file path
        import tempfile
        with tempfile.NamedTemporaryFile(suffix=".gz", delete=False) as f:
            f.write(foreign)
            p = f.name
        try:
            assert P.decompress_file(p) == data
        finally:
            os.unlink(p)

    def test_segment_size_grows_past_table_cap(self):
        assert P._gz_segment_size(10) == P._GZ_SEGMENT
        huge = P._GZ_SEGMENT * P._GZ_MAX_SEGS * 3
        seg = P._gz_segment_size(huge)
        assert huge <= seg * P._GZ_MAX_SEGS

    def test_tfrecord_gzip_roundtrip_cpu(self, tmp_sandbox):
        out = str(tmp_sandbox / "gz")
        data = {"x": np.arange(5000, dtype=np.int64),
                "s": [f"value-{i}" for i in range(5000)]}
        stf.write_tfrecord(data, out, codec="gzip", engine="cpu")
        files = P.list_data_files(out)
        assert all(f.endswith(".tfrecord.gz") for f in files)
        assert P.parse_gz_segments_file(files[0]) is not None
        df = stf.read_tfrecord(out, engine="cpu").sort("x")
        rows = df.collect()
        assert len(rows) == 5000 and rows[17]["s"] == "value-17"


class TestStoredBailout:
    """Incompressible segments re-emit as stored blocks (zstd-style bailout,
    still standard gzip): entropy-coding buys <2% there but costs ~30x on
    decode. Mixed files keep deflate where it pays, per segment."""

    def test_incompressible_goes_stored(self):
        rng = np.random.default_rng(3)
        data = rng.bytes(1 << 20)
        gz = P.compress_bytes(data, "gzip")
        assert len(gz) < len(data) * 1.01  # stored overhead only
        import gzip as _gzip
        assert _gzip.decompress(gz) == data  # foreign readers unaffected

    def test_mixed_file_per_segment_choice(self):
        rng = np.random.default_rng(4)
        data = rng.bytes(200_000) + b"abc" * 100_000 + rng.bytes(200_000)
        gz = P.compress_bytes(data, "gzip")
        import gzip as _gzip
        assert _gzip.decompress(gz) == data
        # the compressible middle must still shrink the whole file
        assert len(gz) < len(data) * 0.95
        # and every segment round-trips through the shared inflate core
        body_off, segs, _, _ = P.parse_gz_segments(gz)
        pos, upos = body_off, 0
        from spark_tfrecord_amd import _native
        for c, u in segs:
            assert _native.host_inflate_segment(gz[pos:pos + c], u) == \
                data[upos:upos + u]
            pos += c
            upos += u

    def test_threshold_disable(self, monkeypatch):
        monkeypatch.setattr(P, "_GZ_STORED_THRESHOLD", 0.0)
        rng = np.random.default_rng(5)
        data = rng.bytes(200_000)
        gz = P.compress_bytes(data, "gzip")
        import gzip as _gzip
        assert _gzip.decompress(gz) == data
