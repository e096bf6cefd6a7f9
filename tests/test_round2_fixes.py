"""Round-2 correctness fixes.

Covers: hidden temp files (a crashed job's leftovers are never listed as
data), Hive-style partition-value escaping (TFRecordIOSuite.scala:140-151
layout contract under Spark's escapePathName), and rejection of 2-D ragged
columns under recordType=Example (TFRecordSerializer.scala:147-180 raises
for types the record type cannot carry — silent drop would be data loss).
"""

import os

import numpy as np
import pyarrow as pa
import pytest

import spark_tfrecord_amd as stf
from spark_tfrecord_amd.io import paths as P


# ---------------------------------------------------------------------------
# Hidden temp files
# ---------------------------------------------------------------------------

class TestTempFileVisibility:
    def test_hidden_tmp_path_is_dotfile(self):
        t = P.hidden_tmp_path("/data/out/part-00000.tfrecord")
        assert os.path.basename(t).startswith(".")
        assert os.path.dirname(t) == "/data/out"

    def test_crashed_job_leftovers_not_read(self, tmp_sandbox):
        out = str(tmp_sandbox / "ds")
        stf.write_tfrecord({"x": np.arange(5, dtype=np.int64)}, out)
        # leftovers in every historical temp style: none may be listed
        for junk in (".part-00009-dead.tfrecord.inprogress",
                     "part-00009-dead.tfrecord.inprogress",       # legacy
                     "part-00009-dead.tfrecord.__tmp.abcd1234",   # legacy
                     ".part-00009-dead.tfrecord.tmp.ffff0000"):
            with open(os.path.join(out, junk), "wb") as f:
                f.write(b"\x00" * 7)  # torn frame header
        files = P.list_data_files(out)
        assert len(files) == 1
        df = stf.read_tfrecord(out)
        assert sorted(r["x"] for r in df.collect()) == list(range(5))

    def test_write_atomic_tmp_is_hidden(self, tmp_sandbox):
        final = str(tmp_sandbox / "d" / "f.tfrecord")
        P.write_file_atomic(b"abc", final)
        assert open(final, "rb").read() == b"abc"
        assert os.listdir(os.path.dirname(final)) == ["f.tfrecord"]

    def test_shard_writer_abort_leaves_no_visible_file(self, tmp_sandbox):
        out = str(tmp_sandbox / "sw")
        os.makedirs(out)
        path = os.path.join(out, "part-00000.tfrecord")
        w = stf.ShardWriter(path, record_type="Example")
        w.write({"x": np.arange(3, dtype=np.int64)})
        # mid-stream: the in-progress file must not be listed as data
        assert P.list_data_files(out) == []
        w.abort()
        assert os.listdir(out) == []


# ---------------------------------------------------------------------------
# Partition-value escaping
# ---------------------------------------------------------------------------

class TestPartitionEscaping:
    def test_escape_unescape_roundtrip(self):
        for s in ("a/b", "k=v", "100%", "a b", "x:y", "plain",
                  "nested/deep/path", "%2F", "tab\tchar", 'q"uote'):
            esc = P.escape_path_name(s)
            assert "/" not in esc and "=" not in esc
            assert P.unescape_path_name(esc) == s

    def test_partitioned_write_read_special_chars(self, tmp_sandbox):
        out = str(tmp_sandbox / "esc")
        vals = ["a/b", "k=v", "100%", "a b", "plain"]
        data = {
            "part": vals * 2,
            "x": np.arange(10, dtype=np.int64),
        }
        stf.write_tfrecord(data, out, partition_by=["part"])
        # layout: every dir component is escaped, no nested dirs from '/'
        dirs = sorted(d for d in os.listdir(out) if d.startswith("part="))
        assert "part=a%2Fb" in dirs
        assert "part=k%3Dv" in dirs
        assert "part=100%25" in dirs
        assert not os.path.isdir(os.path.join(out, "part=a"))
        rows = stf.read_tfrecord(out).collect()
        got = sorted((r["part"], r["x"]) for r in rows)
        want = sorted(zip(vals * 2, range(10)))
        assert got == want

    def test_null_and_empty_partition_values(self, tmp_sandbox):
        out = str(tmp_sandbox / "nulls")
        data = pa.table({
            "part": pa.array(["a", None, ""], type=pa.large_utf8()),
            "x": pa.array([1, 2, 3], type=pa.int64()),
        })
        stf.write_tfrecord(data, out, partition_by=["part"])
        dirs = sorted(d for d in os.listdir(out) if d.startswith("part="))
        assert "part=__HIVE_DEFAULT_PARTITION__" in dirs
        rows = stf.read_tfrecord(out).collect()
        by_x = {r["x"]: r["part"] for r in rows}
        assert by_x[1] == "a"
        assert by_x[2] is None  # Hive default partition reads back as null


# ---------------------------------------------------------------------------
# 2-D ragged columns under recordType=Example must raise, not drop
# ---------------------------------------------------------------------------

def _seq_table():
    return pa.table({
        "mat": pa.array([[[1.0, 2.0], [3.0]]],
                        type=pa.large_list(pa.large_list(pa.float32()))),
    })


def _seq_schema():
    return stf.StructType([
        stf.StructField("mat",
                        stf.ArrayType(stf.ArrayType(stf.FloatType())), True),
    ])


class TestExampleRejectsSeqFields:
    def test_write_example_with_2d_raises(self, tmp_sandbox):
        out = str(tmp_sandbox / "w")
        with pytest.raises(TypeError, match="SequenceExample"):
            stf.write_tfrecord(_seq_table(), out, record_type="Example")

    def test_write_sequence_example_ok(self, tmp_sandbox):
        out = str(tmp_sandbox / "ok")
        stf.write_tfrecord(_seq_table(), out, record_type="SequenceExample")
        rows = stf.read_tfrecord(out, record_type="SequenceExample").collect()
        assert rows[0]["mat"] == [[pytest.approx(1.0), pytest.approx(2.0)],
                                  [pytest.approx(3.0)]]

    def test_read_example_with_2d_schema_raises(self, tmp_sandbox):
        out = str(tmp_sandbox / "r")
        stf.write_tfrecord(_seq_table(), out, record_type="SequenceExample")
        with pytest.raises(TypeError, match="SequenceExample"):
            stf.read_tfrecord(out, schema=_seq_schema(), record_type="Example")

    def test_shard_writer_example_with_2d_raises(self, tmp_sandbox):
        path = str(tmp_sandbox / "part-00000.tfrecord")
        with stf.ShardWriter(path, record_type="Example") as w:
            with pytest.raises(TypeError, match="SequenceExample"):
                w.write(_seq_table())
            w.abort()
