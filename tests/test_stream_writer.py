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
