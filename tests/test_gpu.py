"""GPU-engine tests (gfx950): numerics vs the CPU host codec (the plain
reference implementation of the same ops), CRC kernel behavior, and
end-to-end read/write through engine='gpu'."""

import os

import numpy as np
import pyarrow as pa
import pytest

import spark_tfrecord_amd as stf
from spark_tfrecord_amd.columnar import RecordBatch, column_from_values
from spark_tfrecord_amd.engine import cpu as cpu_engine

pytestmark = pytest.mark.gpu


def _gpu_engine():
    from spark_tfrecord_amd.engine import gpu as gpu_engine
    return gpu_engine


def make_batch(n=257, seed=0):
    rng = np.random.default_rng(seed)
    schema = stf.StructType([
        stf.StructField("id", stf.LongType(), True),
        stf.StructField("f", stf.FloatType(), True),
        stf.StructField("s", stf.StringType(), True),
        stf.StructField("arr", stf.ArrayType(stf.LongType()), True),
        stf.StructField("farr", stf.ArrayType(stf.FloatType()), True),
    ])
    ids = rng.integers(-2**60, 2**60, n)
    fs = rng.random(n).astype(np.float32)
    ss = [f"name-{i}" * (i % 3 + 1) if i % 7 else None for i in range(n)]
    arrs = [list(rng.integers(-100, 100, i % 5)) for i in range(n)]
    farrs = [list(rng.random(i % 8).astype(float)) for i in range(n)]
    cols = [
        column_from_values(ids, stf.LongType(), True, "id"),
        column_from_values(fs, stf.FloatType(), True, "f"),
        column_from_values(ss, stf.StringType(), True, "s"),
        column_from_values(arrs, stf.ArrayType(stf.LongType()), True, "arr"),
        column_from_values(farrs, stf.ArrayType(stf.FloatType()), True, "farr"),
    ]
    return RecordBatch(schema, cols, n)


def assert_batches_equal(a: RecordBatch, b: RecordBatch):
    assert a.num_rows == b.num_rows
    for ca, cb in zip(a.columns, b.columns):
        np.testing.assert_array_equal(np.asarray(ca.presence), np.asarray(cb.presence))
        np.testing.assert_array_equal(np.asarray(ca.row_off), np.asarray(cb.row_off))
        np.testing.assert_array_equal(np.asarray(ca.values), np.asarray(cb.values))
        for attr in ("elem_off", "list_off", "sub_off"):
            va, vb = getattr(ca, attr), getattr(cb, attr)
            assert (va is None) == (vb is None)
            if va is not None:
                np.testing.assert_array_equal(np.asarray(va), np.asarray(vb))


class TestGpuCodec:
    def test_native_has_gpu_kernels(self):
        from spark_tfrecord_amd import _native
        assert _native.HAS_GPU_KERNELS

    def test_encode_matches_cpu(self):
        g = _gpu_engine()
        batch = make_batch()
        cpu_img = cpu_engine.encode_batch(batch, "Example")
        gpu_img = g.encode_batch_from_cpu(batch, "Example")
        assert cpu_img == gpu_img  # byte-for-byte identical framing + payload

    def test_decode_matches_cpu(self):
        g = _gpu_engine()
        batch = make_batch(513, seed=1)
        img = cpu_engine.encode_batch(batch, "Example")
        data = np.frombuffer(img, np.uint8)
        cpu_out = cpu_engine.decode_buffer(data, batch.schema, "Example")
        gpu_out = g.decode_buffer_to_cpu(data, batch.schema, "Example")
        assert_batches_equal(cpu_out, gpu_out)

    def test_gpu_roundtrip_device_only(self):
        g = _gpu_engine()
        batch = make_batch(1000, seed=2)
        dev = g.batch_to_device(batch)
        img_dev = g.encode_device(dev, "Example")
        import torch
        off_np, len_np = [], []
        img_np = img_dev.cpu().numpy()
        from spark_tfrecord_amd import _native
        off, lens = _native.scan_frame_headers(img_np)
        out = g.decode_device(img_dev,
                              torch.as_tensor(off).cuda(),
                              torch.as_tensor(lens).cuda(),
                              batch.schema, "Example", verify_crc=True)
        assert_batches_equal(batch, g.batch_to_host(out))

    def test_crc_kernel_detects_corruption(self):
        g = _gpu_engine()
        import torch
        batch = make_batch(64, seed=3)
        img = bytearray(cpu_engine.encode_batch(batch, "Example"))
        img[20] ^= 0xFF
        with pytest.raises(RuntimeError, match="CRC"):
            g.decode_buffer_to_cpu(np.frombuffer(bytes(img), np.uint8),
                                   batch.schema, "Example")

    def test_sequence_example_roundtrip(self):
        g = _gpu_engine()
        rng = np.random.default_rng(4)
        schema = stf.StructType([
            stf.StructField("ctx", stf.LongType(), True),
            stf.StructField("rag", stf.ArrayType(stf.ArrayType(stf.FloatType())), True),
        ])
        n = 100
        ctx = list(rng.integers(0, 10, n))
        rag = [[list(rng.random(rng.integers(0, 4)).astype(float))
                for _ in range(rng.integers(0, 3))] for _ in range(n)]
        cols = [column_from_values(ctx, stf.LongType(), True, "ctx"),
                column_from_values(rag, schema[1].dataType, True, "rag")]
        batch = RecordBatch(schema, cols, n)
        cpu_img = cpu_engine.encode_batch(batch, "SequenceExample")
        gpu_img = g.encode_batch_from_cpu(batch, "SequenceExample")
        assert cpu_img == gpu_img
        out = g.decode_buffer_to_cpu(np.frombuffer(cpu_img, np.uint8), schema,
                                     "SequenceExample")
        assert_batches_equal(cpu_engine.decode_buffer(
            np.frombuffer(cpu_img, np.uint8), schema, "SequenceExample"), out)

    def test_byte_array_roundtrip(self):
        g = _gpu_engine()
        payloads = [b"\x00\x01", b"", b"abcdef" * 100]
        data = pa.table({"byteArray": pa.array(payloads, type=pa.large_binary())})
        from spark_tfrecord_amd.arrow_interop import table_to_batch
        from spark_tfrecord_amd.infer import byte_array_schema
        batch = table_to_batch(data, byte_array_schema())
        cpu_img = cpu_engine.encode_batch(batch, "ByteArray")
        gpu_img = g.encode_batch_from_cpu(batch, "ByteArray")
        assert cpu_img == gpu_img
        out = g.decode_buffer_to_cpu(np.frombuffer(cpu_img, np.uint8),
                                     byte_array_schema(), "ByteArray")
        assert_batches_equal(cpu_engine.decode_buffer(
            np.frombuffer(cpu_img, np.uint8), byte_array_schema(), "ByteArray"), out)


class TestPipelinedFileIO:
    def test_write_batch_to_file_bytes_match_cpu(self, tmp_path):
        g = _gpu_engine()
        batch = make_batch(3000, seed=6)
        path = str(tmp_path / "p.tfrecord")
        n = g.write_batch_to_file(g.batch_to_device(batch), path, "Example")
        on_disk = open(path, "rb").read()
        assert len(on_disk) == n == os.path.getsize(path)
        assert on_disk == cpu_engine.encode_batch(batch, "Example")

    def test_pipelined_read_matches_cpu(self, tmp_path):
        g = _gpu_engine()
        batch = make_batch(4000, seed=7)
        path = str(tmp_path / "q.tfrecord")
        with open(path, "wb") as f:
            f.write(cpu_engine.encode_batch(batch, "Example"))
        out = g.batch_to_host(g.read_file_to_batch_pipelined(
            path, batch.schema, "Example", verify_crc=True))
        assert_batches_equal(batch, out)

    def test_pipelined_read_multi_slice(self, tmp_path, monkeypatch):
        # force several H2D slices so the guard-band handoff between the
        # sliced frame-candidate launches is exercised on a small file
        g = _gpu_engine()
        monkeypatch.setattr(g, "_READ_SLICE", 64 << 10)  # 64 KiB slices
        batch = make_batch(5000, seed=10)
        path = str(tmp_path / "s.tfrecord")
        with open(path, "wb") as f:
            f.write(cpu_engine.encode_batch(batch, "Example"))
        assert os.path.getsize(path) > 4 * (64 << 10)
        out = g.batch_to_host(g.read_file_to_batch_pipelined(
            path, batch.schema, "Example", verify_crc=True))
        assert_batches_equal(batch, out)

    def test_roundtrip_overwrites_shorter_file(self, tmp_path):
        # in-place mmap reuse must truncate correctly when the file shrinks
        g = _gpu_engine()
        path = str(tmp_path / "r.tfrecord")
        big = make_batch(2000, seed=8)
        small = make_batch(50, seed=9)
        g.write_batch_to_file(g.batch_to_device(big), path, "Example")
        g.write_batch_to_file(g.batch_to_device(small), path, "Example")
        assert open(path, "rb").read() == cpu_engine.encode_batch(small, "Example")
        out = g.batch_to_host(g.read_file_to_batch_pipelined(
            path, small.schema, "Example"))
        assert_batches_equal(small, out)


class TestGpuInference:
    def _codes_both(self, batch, record_type):
        import torch
        from spark_tfrecord_amd import _native
        from spark_tfrecord_amd.infer import infer_codes_from_buffer
        g = _gpu_engine()
        img = cpu_engine.encode_batch(batch, record_type)
        data = np.frombuffer(img, np.uint8)
        off, lens = _native.scan_frames(data, False)
        cpu_codes = infer_codes_from_buffer(data, off, lens, record_type)
        dev = torch.as_tensor(np.ascontiguousarray(data).copy()).cuda()
        gpu_codes = g.infer_codes_device(
            dev, torch.as_tensor(off).cuda(), torch.as_tensor(lens).cuda(),
            record_type)
        return cpu_codes, gpu_codes

    def test_infer_example_matches_host(self):
        cpu_codes, gpu_codes = self._codes_both(make_batch(777, seed=11),
                                                "Example")
        assert cpu_codes == gpu_codes

    def test_infer_sequence_matches_host(self):
        rng = np.random.default_rng(12)
        schema = stf.StructType([
            stf.StructField("ctx", stf.LongType(), True),
            stf.StructField("rag", stf.ArrayType(stf.ArrayType(stf.FloatType())), True),
        ])
        n = 300
        ctx = list(rng.integers(0, 10, n))
        rag = [[list(rng.random(rng.integers(0, 4)).astype(float))
                for _ in range(rng.integers(0, 3))] for _ in range(n)]
        cols = [column_from_values(ctx, stf.LongType(), True, "ctx"),
                column_from_values(rag, schema[1].dataType, True, "rag")]
        batch = RecordBatch(schema, cols, n)
        cpu_codes, gpu_codes = self._codes_both(batch, "SequenceExample")
        assert cpu_codes == gpu_codes

    def test_read_with_gpu_inference(self, tmp_sandbox):
        out = str(tmp_sandbox / "gi")
        data = {"a": np.arange(100, dtype=np.int64),
                "b": [[1.5, 2.5]] * 100,
                "c": [f"s{i}" for i in range(100)]}
        stf.write_tfrecord(data, out, engine="gpu")
        df = stf.read_tfrecord(out, engine="gpu").sort("a")  # schema inferred
        rows = df.collect()
        assert rows[3]["a"] == 3 and rows[3]["b"] == [1.5, 2.5]
        assert rows[3]["c"] == "s3"


class TestGpuRobustness:
    """Hostile inputs must raise cleanly — never crash a kernel or hang."""

    def test_random_garbage_raises(self, tmp_path):
        g = _gpu_engine()
        rng = np.random.default_rng(0)
        path = str(tmp_path / "garbage.tfrecord")
        with open(path, "wb") as f:
            f.write(rng.bytes(1 << 20))
        schema = stf.StructType([stf.StructField("x", stf.LongType(), True)])
        with pytest.raises(RuntimeError, match="corrupt TFRecord"):
            g.read_file_to_batch_pipelined(path, schema, "Example")

    def test_truncated_file_raises(self, tmp_path):
        g = _gpu_engine()
        batch = make_batch(500, seed=1)
        img = cpu_engine.encode_batch(batch, "Example")
        path = str(tmp_path / "trunc.tfrecord")
        with open(path, "wb") as f:
            f.write(img[:-7])  # chop mid-frame
        with pytest.raises(RuntimeError, match="corrupt TFRecord"):
            g.read_file_to_batch_pipelined(path, batch.schema, "Example")

    def test_wrong_schema_kind_raises(self, tmp_path):
        g = _gpu_engine()
        batch = make_batch(100, seed=2)
        path = str(tmp_path / "k.tfrecord")
        with open(path, "wb") as f:
            f.write(cpu_engine.encode_batch(batch, "Example"))
        wrong = stf.StructType([stf.StructField("id", stf.StringType(), True)])
        with pytest.raises(RuntimeError):
            g.read_file_to_batch_pipelined(path, wrong, "Example")

    def test_count_and_projection_gpu(self, tmp_sandbox):
        out = str(tmp_sandbox / "cp")
        stf.write_tfrecord({"a": np.arange(2000, dtype=np.int64),
                            "b": np.arange(2000, dtype=np.float32)},
                           out, engine="gpu", num_shards=2)
        assert stf.count_tfrecord(out, engine="gpu") == 2000
        df = stf.read_tfrecord(out, engine="gpu", columns=["b"])
        assert df.columns == ["b"] and df.count() == 2000


class TestGpuRandomized:
    """Randomized numerics sweep: many random batches, GPU kernels vs the
    plain host codec (the CPU reference implementation of the same ops)."""

    @pytest.mark.parametrize("seed", range(8))
    def test_random_batches_encode_decode(self, seed):
        g = _gpu_engine()
        rng = np.random.default_rng(1000 + seed)
        n = int(rng.integers(1, 400))
        batch = make_batch(n, seed=seed)
        cpu_img = cpu_engine.encode_batch(batch, "Example")
        assert g.encode_batch_from_cpu(batch, "Example") == cpu_img
        data = np.frombuffer(cpu_img, np.uint8)
        assert_batches_equal(
            cpu_engine.decode_buffer(data, batch.schema, "Example"),
            g.decode_buffer_to_cpu(data, batch.schema, "Example"))

    def test_single_record_file(self, tmp_path):
        g = _gpu_engine()
        batch = make_batch(1, seed=42)
        path = str(tmp_path / "one.tfrecord")
        g.write_batch_to_file(g.batch_to_device(batch), path, "Example")
        out = g.batch_to_host(g.read_file_to_batch_pipelined(
            path, batch.schema, "Example"))
        assert_batches_equal(batch, out)


class TestTorchDatasetGpu:
    def test_device_resident_stream(self, tmp_sandbox):
        import torch

        from spark_tfrecord_amd.torch_data import TFRecordIterableDataset

        out = str(tmp_sandbox / "tds")
        rng = np.random.default_rng(0)
        stf.write_tfrecord({"uid": np.arange(3000, dtype=np.int64),
                            "v": rng.random(3000).astype(np.float32)},
                           out, engine="gpu", num_shards=3)
        ds = TFRecordIterableDataset(out, batch_rows=512, engine="gpu")
        total = 0
        seen = []
        for b in ds:
            assert b["uid"].is_cuda and b["v"].is_cuda  # stays on device
            total += int(b["_num_rows"])
            seen.append(b["uid"].cpu())
        assert total == 3000
        got = np.sort(torch.cat(seen).numpy())
        np.testing.assert_array_equal(got, np.arange(3000))


class TestGpuEndToEnd:
    def test_write_read_files_gpu_engine(self, tmp_sandbox):
        out = str(tmp_sandbox / "g")
        rng = np.random.default_rng(5)
        data = {
            "id": np.arange(5000, dtype=np.int64),
            "v": rng.random(5000).astype(np.float32),
            "tags": [[f"t{i % 13}", f"u{i % 7}"] for i in range(5000)],
        }
        stf.write_tfrecord(data, out, engine="gpu")
        df = stf.read_tfrecord(out, engine="gpu").sort("id")
        rows = df.collect()
        assert len(rows) == 5000
        assert rows[17]["tags"] == ["t4", "u3"]
        np.testing.assert_allclose([r["v"] for r in rows[:10]], data["v"][:10],
                                   rtol=1e-6)

    def test_auto_engine_uses_gpu(self, tmp_sandbox):
        from spark_tfrecord_amd.engine import resolve_engine
        assert resolve_engine("auto") == "gpu"

    def test_gpu_partitioned_write(self, tmp_sandbox):
        out = str(tmp_sandbox / "p")
        data = {"part": np.array([1, 1, 2, 2], np.int64),
                "x": np.arange(4, dtype=np.int64)}
        stf.write_tfrecord(data, out, partition_by=["part"], engine="gpu")
        assert sorted(d for d in os.listdir(out) if d.startswith("part=")) == \
            ["part=1", "part=2"]
        df = stf.read_tfrecord(out, engine="gpu").sort("x")
        assert [r["part"] for r in df.collect()] == [1, 1, 2, 2]

    def test_gpu_multi_column_partition(self, tmp_sandbox):
        out = str(tmp_sandbox / "mc")
        data = {"a": np.array([1, 1, 2, 2, 1], np.int64),
                "b": ["x", "y", "x", "x", "y"],
                "v": np.arange(5, dtype=np.int64)}
        stf.write_tfrecord(data, out, partition_by=["a", "b"], engine="gpu")
        dirs = sorted(f"{d}/{s}" for d in os.listdir(out) if d.startswith("a=")
                      for s in os.listdir(os.path.join(out, d)))
        assert dirs == ["a=1/b=x", "a=1/b=y", "a=2/b=x"]
        df = stf.read_tfrecord(out, engine="gpu").sort("v")
        rows = df.collect()
        assert [(r["a"], r["b"], r["v"]) for r in rows] == \
            [(1, "x", 0), (1, "y", 1), (2, "x", 2), (2, "x", 3), (1, "y", 4)]


class TestShardWriterGpu:
    def test_stream_writer_gpu_engine(self, tmp_sandbox):
        p = str(tmp_sandbox / "swg" / "part-00000.tfrecord")
        with stf.ShardWriter(p, engine="gpu") as w:
            for k in range(4):
                w.write({"x": np.arange(k * 100, k * 100 + 100,
                                        dtype=np.int64)})
        df = stf.read_tfrecord(p, engine="gpu").sort("x")
        assert df.count() == 400
        assert [r["x"] for r in df.collect()[:3]] == [0, 1, 2]


class TestWaveRecords:
    """Large records go one-per-wavefront (cooperative copy + GF(2)-combined
    chunk CRCs) — outputs must stay byte-identical to the host codec."""

    def test_big_blob_bytearray_roundtrip(self, tmp_path):
        g = _gpu_engine()
        rng = np.random.default_rng(0)
        payloads = [rng.bytes(int(rng.integers(1, 400_000))) for _ in range(40)]
        data = pa.table({"byteArray": pa.array(payloads, type=pa.large_binary())})
        from spark_tfrecord_amd.arrow_interop import table_to_batch
        from spark_tfrecord_amd.infer import byte_array_schema
        batch = table_to_batch(data, byte_array_schema())
        cpu_img = cpu_engine.encode_batch(batch, "ByteArray")
        assert g.encode_batch_from_cpu(batch, "ByteArray") == cpu_img
        out = g.decode_buffer_to_cpu(np.frombuffer(cpu_img, np.uint8),
                                     byte_array_schema(), "ByteArray")
        assert_batches_equal(cpu_engine.decode_buffer(
            np.frombuffer(cpu_img, np.uint8), byte_array_schema(), "ByteArray"),
            out)

    def test_big_record_example_roundtrip_and_crc(self, tmp_path):
        g = _gpu_engine()
        rng = np.random.default_rng(1)
        n = 30
        big = [("B" * int(rng.integers(30_000, 120_000))) for _ in range(n)]
        schema = stf.StructType([stf.StructField("blob", stf.StringType(), True)])
        batch = RecordBatch(schema, [column_from_values(big, stf.StringType(),
                                                        True, "blob")], n)
        img = cpu_engine.encode_batch(batch, "Example")
        out = g.decode_buffer_to_cpu(np.frombuffer(img, np.uint8), schema,
                                     "Example", verify_crc=True)
        assert_batches_equal(cpu_engine.decode_buffer(
            np.frombuffer(img, np.uint8), schema, "Example"), out)
        # corruption in a big record must still be caught (wave CRC path)
        bad = bytearray(img)
        bad[len(bad) // 2] ^= 0x40
        with pytest.raises(RuntimeError, match="CRC"):
            g.decode_buffer_to_cpu(np.frombuffer(bytes(bad), np.uint8),
                                   schema, "Example", verify_crc=True)


class TestWideSchemaGpu:
    def test_100_fields_gpu_roundtrip(self, tmp_sandbox):
        out = str(tmp_sandbox / "wide")
        data = {f"f{i:03d}": np.arange(200, dtype=np.int64) + i
                for i in range(100)}
        stf.write_tfrecord(data, out, engine="gpu")
        df = stf.read_tfrecord(out, engine="gpu").sort("f000")
        assert len(df.columns) == 100 and df.count() == 200
        assert df.collect()[5]["f099"] == 104


class TestStreamOrderingStress:
    def test_mmap_cache_dma_churn(self, tmp_path):
        """Size churn over two cached paths: every write retruncates and
        re-registers the inode mapping while reads DMA from it on the side
        streams — the stress shape for the drop-before-ftruncate ordering
        in file_mmap_pinned (kernels.hip mapping cache)."""
        g = _gpu_engine()
        rng = np.random.default_rng(42)
        paths = [str(tmp_path / f"st{i}.tfrecord") for i in range(2)]
        for it in range(25):
            n = int(rng.integers(1, 4000))
            b = make_batch(n, seed=it)
            p = paths[it % 2]
            g.write_batch_to_file(g.batch_to_device(b), p, "Example")
            out = g.batch_to_host(g.read_file_to_batch_pipelined(
                p, b.schema, "Example", verify_crc=True))
            assert_batches_equal(b, out)

    def test_interleaved_raw_and_batch_io(self, tmp_path):
        """device_to_file / read_file_to_device interleaved with the framed
        pipeline on the same two DMA streams must not reorder."""
        import torch
        g = _gpu_engine()
        raw_path = str(tmp_path / "raw.bin")
        rec_path = str(tmp_path / "rec.tfrecord")
        batch = make_batch(1000, seed=3)
        for it in range(10):
            raw = torch.arange(
                it + 1, dtype=torch.float32, device="cuda").view(torch.uint8)
            g.device_to_file(raw.contiguous(), raw_path)
            g.write_batch_to_file(g.batch_to_device(batch), rec_path, "Example")
            back = g.read_file_to_device(raw_path)
            assert torch.equal(back.cpu(), raw.cpu())
            out = g.batch_to_host(g.read_file_to_batch_pipelined(
                rec_path, batch.schema, "Example"))
            assert_batches_equal(batch, out)


class TestInferenceLimits:
    def test_5000_distinct_names_grows_table(self):
        """>1024 names per table half forces the ERR_OVERFLOW retry with a
        grown table (2048-slot hard cap lifted)."""
        import torch
        g = _gpu_engine()
        F, R = 5000, 2
        schema = stf.StructType([
            stf.StructField(f"feat_{i:05d}", stf.LongType(), True)
            for i in range(F)])
        cols = [column_from_values(np.arange(R, dtype=np.int64),
                                   stf.LongType(), True, f"feat_{i:05d}")
                for i in range(F)]
        img = cpu_engine.encode_batch(RecordBatch(schema, cols, R), "Example")
        data = torch.frombuffer(bytearray(img), dtype=torch.uint8).cuda()
        off, lens = g.scan_frames_device(data)
        codes = g.infer_codes_device(data, off, lens, "Example")
        assert len(codes) == F
        assert set(codes.values()) == {1}  # all long scalars
        assert sorted(codes)[0] == "feat_00000"


class TestErrorIndices:
    def test_kind_mismatch_reports_record_index(self):
        """Record 3 carries a float where the schema says int64: the decode
        error must name a record index."""
        import torch
        schema_l = stf.StructType([stf.StructField("x", stf.LongType(), True)])
        schema_f = stf.StructType([stf.StructField("x", stf.FloatType(), True)])
        good = cpu_engine.encode_batch(RecordBatch(schema_l, [
            column_from_values(np.arange(3, dtype=np.int64), stf.LongType(),
                               True, "x")], 3), "Example")
        bad = cpu_engine.encode_batch(RecordBatch(schema_f, [
            column_from_values(np.array([1.5], np.float32), stf.FloatType(),
                               True, "x")], 1), "Example")
        g = _gpu_engine()
        with pytest.raises(RuntimeError, match=r"record 3"):
            g.decode_buffer_to_cpu(np.frombuffer(good + bad, np.uint8),
                                   schema_l, "Example", verify_crc=False)


class TestWaveProtoPath:
    """Wave-cooperative scan/extract/emit for >8KB records (round-2 roadmap
    item): numerics vs the CPU host codec, byte-exact emit, all value kinds
    including negative int64s (10-byte varints) through the parallel
    packed-varint decode."""

    def _big_example_batch(self, n=48, seed=0):
        rng = np.random.default_rng(seed)
        schema = stf.StructType([
            stf.StructField("ints", stf.ArrayType(stf.LongType()), True),
            stf.StructField("floats", stf.ArrayType(stf.FloatType()), True),
            stf.StructField("blob", stf.StringType(), True),
            stf.StructField("small", stf.LongType(), True),
        ])
        ints = [list(rng.integers(-2**62, 2**62,
                                  int(rng.integers(800, 1500))))
                for _ in range(n)]
        floats = [list(rng.random(int(rng.integers(500, 900))).astype(float))
                  for _ in range(n)]
        blobs = ["x" * int(rng.integers(4000, 9000)) if i % 5 else None
                 for i in range(n)]
        small = rng.integers(0, 100, n)
        cols = [
            column_from_values(ints, stf.ArrayType(stf.LongType()), True, "ints"),
            column_from_values(floats, stf.ArrayType(stf.FloatType()), True, "floats"),
            column_from_values(blobs, stf.StringType(), True, "blob"),
            column_from_values(small, stf.LongType(), True, "small"),
        ]
        return RecordBatch(schema, cols, n)

    def test_wave_decode_matches_cpu(self):
        g = _gpu_engine()
        batch = self._big_example_batch()
        img = cpu_engine.encode_batch(batch, "Example")
        assert len(img) // batch.num_rows > 8 << 10  # wave path engaged
        out = g.decode_buffer_to_cpu(np.frombuffer(img, np.uint8),
                                     batch.schema, "Example", verify_crc=True)
        assert_batches_equal(batch, out)

    def test_wave_emit_bytes_match_cpu(self, tmp_path):
        g = _gpu_engine()
        batch = self._big_example_batch(seed=3)
        path = str(tmp_path / "wave.tfrecord")
        g.write_batch_to_file(g.batch_to_device(batch), path, "Example")
        assert open(path, "rb").read() == cpu_engine.encode_batch(batch, "Example")

    def test_wave_sequence_example(self):
        g = _gpu_engine()
        rng = np.random.default_rng(5)
        n = 24
        schema = stf.StructType([
            stf.StructField("ctx", stf.LongType(), True),
            stf.StructField("seq",
                            stf.ArrayType(stf.ArrayType(stf.FloatType())), True),
        ])
        seqs = [[list(rng.random(int(rng.integers(200, 400))).astype(float))
                 for _ in range(int(rng.integers(8, 16)))] for _ in range(n)]
        cols = [
            column_from_values(np.arange(n, dtype=np.int64), stf.LongType(),
                               True, "ctx"),
            column_from_values(
                seqs, stf.ArrayType(stf.ArrayType(stf.FloatType())), True, "seq"),
        ]
        batch = RecordBatch(schema, cols, n)
        img = cpu_engine.encode_batch(batch, "SequenceExample")
        assert len(img) // n > 8 << 10
        out = g.decode_buffer_to_cpu(np.frombuffer(img, np.uint8), schema,
                                     "SequenceExample", verify_crc=True)
        assert_batches_equal(batch, out)
        # emit side: byte-exact vs host
        dev_img = g.encode_device(g.batch_to_device(batch), "SequenceExample")
        assert g.device_to_bytes(dev_img) == img

    def test_wave_crc_detects_corruption(self):
        g = _gpu_engine()
        batch = self._big_example_batch(n=20, seed=9)
        img = bytearray(cpu_engine.encode_batch(batch, "Example"))
        img[len(img) * 2 // 3] ^= 0x01
        with pytest.raises(RuntimeError, match="CRC"):
            g.decode_buffer_to_cpu(np.frombuffer(bytes(img), np.uint8),
                                   batch.schema, "Example", verify_crc=True)


class TestDeviceInflate:
    """csrc/hip/inflate.hip: one-segment-per-lane DEFLATE decode of our
    segment-table gzip. Numerics vs host zlib; stored/fixed/dynamic blocks;
    foreign-gzip and corrupt-stream fallbacks."""

    def _gz_file(self, tmp_path, data, name="t.tfrecord.gz"):
        from spark_tfrecord_amd.io import paths as P
        p = str(tmp_path / name)
        with open(p, "wb") as f:
            f.write(P.compress_bytes(data, "gzip"))
        return p

    def test_inflate_matches_zlib_compressible(self, tmp_path):
        from spark_tfrecord_amd.io import paths as P
        g = _gpu_engine()
        rng = np.random.default_rng(0)
        # highly compressible -> dynamic-Huffman blocks, long matches
        data = bytes(rng.integers(65, 75, 3 * P._GZ_SEGMENT + 12345)
                     .astype(np.uint8))
        p = self._gz_file(tmp_path, data)
        dev = g.read_gzip_file_to_device(p)
        assert dev is not None
        assert bytes(dev.cpu().numpy().tobytes()) == data

    def test_inflate_matches_zlib_incompressible(self, tmp_path):
        from spark_tfrecord_amd.io import paths as P
        g = _gpu_engine()
        rng = np.random.default_rng(1)
        # random bytes -> stored blocks inside the deflate stream
        data = rng.bytes(2 * P._GZ_SEGMENT + 999)
        p = self._gz_file(tmp_path, data)
        dev = g.read_gzip_file_to_device(p)
        assert dev is not None
        assert bytes(dev.cpu().numpy().tobytes()) == data

    def test_inflate_mixed_and_tiny(self, tmp_path):
        from spark_tfrecord_amd.io import paths as P
        g = _gpu_engine()
        rng = np.random.default_rng(2)
        pieces = [b"", b"a", b"ab" * 50000, rng.bytes(70000),
                  bytes(rng.integers(97, 99, P._GZ_SEGMENT).astype(np.uint8))]
        for i, data in enumerate(pieces):
            p = self._gz_file(tmp_path, data, f"m{i}.tfrecord.gz")
            dev = g.read_gzip_file_to_device(p)
            assert dev is not None, i
            assert bytes(dev.cpu().numpy().tobytes()) == data, i

    def test_gpu_read_gzip_dataset(self, tmp_sandbox):
        out = str(tmp_sandbox / "gzds")
        data = {"x": np.arange(30000, dtype=np.int64),
                "s": [f"value-{i}" for i in range(30000)]}
        stf.write_tfrecord(data, out, codec="gzip", num_shards=4)
        df = stf.read_tfrecord(out, engine="gpu").sort("x")
        rows = df.collect()
        assert len(rows) == 30000 and rows[123]["s"] == "value-123"

    def test_gpu_read_bytearray_gzip(self, tmp_sandbox):
        import pyarrow as pa
        out = str(tmp_sandbox / "gzba")
        rng = np.random.default_rng(5)
        payloads = [rng.bytes(200) for _ in range(5000)]
        t = pa.table({"byteArray": pa.array(payloads, type=pa.large_binary())})
        stf.write_tfrecord(t, out, record_type="ByteArray", codec="gzip",
                           num_shards=8)
        df = stf.read_tfrecord(out, record_type="ByteArray", engine="gpu")
        got = [r["byteArray"] for r in df.collect()]
        assert sorted(got) == sorted(payloads)

    def test_foreign_gzip_falls_back_to_host(self, tmp_sandbox):
        import gzip as _gzip
        out = str(tmp_sandbox / "foreign")
        stf.write_tfrecord({"x": np.arange(100, dtype=np.int64)}, out,
                           engine="cpu")
        from spark_tfrecord_amd.io import paths as P
        src = P.list_data_files(out)[0]
        raw = open(src, "rb").read()
        os.unlink(src)
        with open(src + ".gz", "wb") as f:
            f.write(_gzip.compress(raw, 6))  # table-less foreign gzip
        g = _gpu_engine()
        assert g.read_gzip_file_to_device(src + ".gz") is None
        df = stf.read_tfrecord(out, engine="gpu").sort("x")
        assert [r["x"] for r in df.collect()] == list(range(100))

    def test_corrupt_gzip_body_raises(self, tmp_sandbox):
        out = str(tmp_sandbox / "corrupt")
        stf.write_tfrecord({"x": np.arange(20000, dtype=np.int64)}, out,
                           codec="gzip")
        from spark_tfrecord_amd.io import paths as P
        src = P.list_data_files(out)[0]
        raw = bytearray(open(src, "rb").read())
        raw[len(raw) // 2] ^= 0xFF  # flip inside the compressed body
        with open(src, "wb") as f:
            f.write(bytes(raw))
        with pytest.raises(Exception):
            stf.read_tfrecord(out, engine="gpu").collect()


class TestDeferredInference:
    """Schema-less GPU reads fuse inference into the group pipeline; the
    semantics must stay exactly the reference's (first non-empty file
    decides; columns project; partition dirs still discovered)."""

    def test_schemaless_read_matches_explicit(self, tmp_sandbox):
        out = str(tmp_sandbox / "d")
        data = {"a": np.arange(1000, dtype=np.int64),
                "b": [[float(i), i / 3] for i in range(1000)],
                "s": [f"v{i}" for i in range(1000)]}
        stf.write_tfrecord(data, out, num_shards=3)
        got = stf.read_tfrecord(out, engine="gpu").sort("a").collect()
        assert len(got) == 1000
        assert got[5]["s"] == "v5"
        assert got[7]["b"] == [pytest.approx(7.0), pytest.approx(7 / 3)]

    def test_schemaless_with_columns_projection(self, tmp_sandbox):
        out = str(tmp_sandbox / "p")
        stf.write_tfrecord({"a": np.arange(50, dtype=np.int64),
                            "b": np.arange(50, dtype=np.int64) * 2}, out)
        df = stf.read_tfrecord(out, engine="gpu", columns=["b"])
        assert df.columns == ["b"]
        assert sorted(r["b"] for r in df.collect()) == [2 * i for i in range(50)]
        with pytest.raises(KeyError):
            stf.read_tfrecord(out, engine="gpu", columns=["nope"])

    def test_schemaless_partitioned(self, tmp_sandbox):
        out = str(tmp_sandbox / "part")
        stf.write_tfrecord({"p": ["x", "y"] * 10,
                            "v": np.arange(20, dtype=np.int64)}, out,
                           partition_by=["p"])
        rows = stf.read_tfrecord(out, engine="gpu").sort("v").collect()
        assert len(rows) == 20
        assert rows[0]["p"] == "x" and rows[1]["p"] == "y"

    def test_first_file_empty_fallback(self, tmp_sandbox):
        out = str(tmp_sandbox / "empty_first")
        stf.write_tfrecord({"x": np.arange(5, dtype=np.int64)}, out)
        # an empty part file that sorts FIRST: the fused path must fall
        # through to the non-empty file for the schema
        open(os.path.join(out, "part-00000-aaa.tfrecord"), "wb").close()
        rows = stf.read_tfrecord(out, engine="gpu").collect()
        assert sorted(r["x"] for r in rows) == list(range(5))

    def test_all_files_empty_raises(self, tmp_sandbox):
        out = str(tmp_sandbox / "all_empty")
        os.makedirs(out)
        open(os.path.join(out, "part-00000-aaa.tfrecord"), "wb").close()
        with pytest.raises(ValueError, match="no non-empty"):
            stf.read_tfrecord(out, engine="gpu")


class TestMixedDatasetRead:
    def test_uncompressed_plus_gzip_plus_foreign(self, tmp_sandbox):
        """One dataset mixing uncompressed parts (device DMA), our gzip
        (device inflate) and a foreign table-less gzip (host fallback):
        the reader must merge all three transparently."""
        import gzip as _gzip

        from spark_tfrecord_amd.io import paths as P

        out = str(tmp_sandbox / "mix")
        stf.write_tfrecord({"x": np.arange(0, 100, dtype=np.int64)}, out,
                           num_shards=2)
        stf.write_tfrecord({"x": np.arange(100, 200, dtype=np.int64)}, out,
                           codec="gzip", mode="append", shard_offset=2,
                           write_success=False)
        # foreign gzip: raw frames compressed by the gzip module (no table)
        from spark_tfrecord_amd.engine import cpu as cpu_engine
        from spark_tfrecord_amd.columnar import RecordBatch, column_from_values
        schema = stf.StructType([stf.StructField("x", stf.LongType(), True)])
        b = RecordBatch(schema, [column_from_values(
            np.arange(200, 300, dtype=np.int64), stf.LongType(), True, "x")], 100)
        raw = cpu_engine.encode_batch(b, "Example")
        with open(os.path.join(out, "part-00009-foreign.tfrecord.gz"), "wb") as f:
            f.write(_gzip.compress(raw, 6))
        df = stf.read_tfrecord(out, engine="gpu")
        assert sorted(r["x"] for r in df.collect()) == list(range(300))


class TestGzCountValidate:
    def test_count_and_validate_gzip_on_device(self, tmp_sandbox):
        from spark_tfrecord_amd.io.reader import count_tfrecord
        from spark_tfrecord_amd.io.validate import validate_tfrecord

        out = str(tmp_sandbox / "gzcv")
        stf.write_tfrecord({"x": np.arange(5000, dtype=np.int64)}, out,
                           codec="gzip", num_shards=3)
        assert count_tfrecord(out, engine="gpu") == 5000
        rep = validate_tfrecord(out, engine="gpu")
        assert rep.ok and rep.records == 5000
        # corrupt one compressed body: validation must flag that file
        from spark_tfrecord_amd.io import paths as P
        f = P.list_data_files(out)[0]
        raw = bytearray(open(f, "rb").read())
        raw[len(raw) // 2] ^= 0xFF
        open(f, "wb").write(bytes(raw))
        rep = validate_tfrecord(out, engine="gpu")
        assert not rep.ok


class TestPipelinedWriteGPU:
    def test_chunked_single_shard_matches_oneshot(self, tmp_sandbox, monkeypatch):
        """Default API write (num_shards=1) streams chunk appends while the
        next chunk encodes on the GPU; the file content must be identical
        to the one-shot encode (TFRecord frames are concatenable)."""
        from spark_tfrecord_amd.io import writer as W
        rng = np.random.default_rng(11)
        rows = 50_000
        t = pa.table({
            "id": np.arange(rows, dtype=np.int64),
            "v": rng.random(rows).astype(np.float32),
            "s": pa.array([f"s{i%97}" for i in range(rows)]),
        })
        monkeypatch.setattr(W, "_PIPE_CHUNK_ROWS", 7_000)
        a = str(tmp_sandbox / "gpu_chunked")
        stf.write_tfrecord(t, a, engine="gpu", job_id="jgpu")
        monkeypatch.setattr(W, "_PIPE_CHUNK_ROWS", 10**9)
        b = str(tmp_sandbox / "gpu_oneshot")
        stf.write_tfrecord(t, b, engine="gpu", job_id="jgpu")
        fa = sorted(f for f in os.listdir(a) if f != "_SUCCESS")
        fb = sorted(f for f in os.listdir(b) if f != "_SUCCESS")
        assert fa == fb
        with open(os.path.join(a, fa[0]), "rb") as f1, \
                open(os.path.join(b, fb[0]), "rb") as f2:
            assert f1.read() == f2.read()
        df = stf.read_tfrecord(a, engine="gpu")
        assert len(df.collect()) == rows


class TestStreamedShardDeviceRead:
    def test_streamed_gzip_shard_inflates_on_device(self, tmp_sandbox):
        """A ShardWriter gzip shard carries the FEXTRA segment table, so the
        GPU read path must take the device-inflate branch (gz_device_meta
        not None) and decode identically."""
        from spark_tfrecord_amd.engine import gpu as gpu_engine
        d = str(tmp_sandbox / "swgz")
        os.makedirs(d, exist_ok=True)
        p = os.path.join(d, "part-00000.tfrecord.gz")
        with stf.ShardWriter(p, record_type="Example", codec="gzip",
                             engine="cpu") as w:
            for k in range(4):
                w.write({"x": np.arange(k * 30_000, (k + 1) * 30_000,
                                        dtype=np.int64)})
        assert gpu_engine.gz_device_meta(p) is not None
        df = stf.read_tfrecord(d, engine="gpu")
        assert sorted(r["x"] for r in df.collect()) == list(range(120_000))


class TestGpuSaveModesAndEscaping:
    def test_append_mode_gpu(self, tmp_sandbox):
        out = str(tmp_sandbox / "ap")
        stf.write_tfrecord({"x": np.arange(5000, dtype=np.int64)}, out,
                           engine="gpu")
        stf.write_tfrecord({"x": np.arange(5000, 9000, dtype=np.int64)}, out,
                           engine="gpu", mode="append")
        df = stf.read_tfrecord(out, engine="gpu")
        assert sorted(r["x"] for r in df.collect()) == list(range(9000))

    def test_overwrite_and_ignore_gpu(self, tmp_sandbox):
        out = str(tmp_sandbox / "ow")
        stf.write_tfrecord({"x": np.arange(100, dtype=np.int64)}, out,
                           engine="gpu")
        stf.write_tfrecord({"x": np.arange(50, dtype=np.int64)}, out,
                           engine="gpu", mode="overwrite")
        assert stf.read_tfrecord(out, engine="gpu").count() == 50
        stf.write_tfrecord({"x": np.arange(7, dtype=np.int64)}, out,
                           engine="gpu", mode="ignore")
        assert stf.read_tfrecord(out, engine="gpu").count() == 50

    def test_partition_value_escaping_gpu(self, tmp_sandbox):
        """Hive-style %XX escaping round-trips through the GPU write and
        read paths (values with '/', '=', '%', space)."""
        out = str(tmp_sandbox / "esc")
        vals = ["a/b", "k=v", "100%", "with space", None]
        rows = 5000
        t = pa.table({
            "id": np.arange(rows, dtype=np.int64),
            "p": pa.array([vals[i % len(vals)] for i in range(rows)]),
        })
        stf.write_tfrecord(t, out, engine="gpu", partition_by=["p"])
        df = stf.read_tfrecord(out, engine="gpu")
        got = df.to_arrow_table().to_pylist()
        assert len(got) == rows
        for r in got:
            want = vals[r["id"] % len(vals)]
            assert r["p"] == want

    def test_stored_bailout_gzip_device_read(self, tmp_sandbox):
        """Truly random ByteArray payloads make every segment incompressible;
        the writer emits stored blocks and the device inflater streams them
        at copy speed — content must round-trip exactly."""
        rng = np.random.default_rng(33)
        rows = 30_000
        payloads = [rng.bytes(500) for _ in range(rows)]
        t = pa.table({"byteArray": pa.array(payloads, type=pa.large_binary())})
        out = str(tmp_sandbox / "stored")
        stf.write_tfrecord(t, out, record_type="ByteArray", codec="gzip",
                           num_shards=4, engine="cpu")
        # the dataset must actually be stored-block dominated
        from spark_tfrecord_amd.io import paths as P
        total_gz = sum(os.path.getsize(f) for f in P.list_data_files(out))
        assert total_gz > rows * 500  # ratio >= 1: stored
        df = stf.read_tfrecord(out, record_type="ByteArray", engine="gpu")
        got = sorted(df.to_arrow_table().column("byteArray").to_pylist())
        assert got == sorted(payloads)


class TestThreadConcurrency:
    def test_concurrent_writers_and_readers(self, tmp_sandbox):
        """Two user threads driving full API write+read round-trips
        concurrently must not corrupt each other (pinned staging pools are
        thread-local; kernel launches serialize on the shared stream)."""
        import threading
        errs = []

        def work(tid):
            try:
                rng = np.random.default_rng(tid)
                for it in range(6):
                    rows = 120_000 + 1000 * tid
                    d = str(tmp_sandbox / f"thr{tid}_{it % 2}")
                    vals = rng.integers(0, 2**60, rows)
                    stf.write_tfrecord({"x": vals}, d, engine="gpu",
                                       mode="overwrite")
                    got = stf.read_tfrecord(d, engine="gpu") \
                        .to_arrow_table().column("x").to_numpy()
                    assert np.array_equal(np.sort(got), np.sort(vals)), \
                        f"thread {tid} iter {it} corrupted"
            except BaseException as e:  # noqa: BLE001
                errs.append(e)

        ts = [threading.Thread(target=work, args=(t,)) for t in (1, 2)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(120)
        assert not errs, errs


class TestDuplicateMapKeysGpu:
    def test_last_entry_wins_fused_scan(self, tmp_sandbox):
        """The fused-CRC scan path must apply the same last-entry-wins reset
        as the two-pass form for duplicate Features-map keys."""
        from tests.test_serde import _raw_example_with_dup_keys
        d = str(tmp_sandbox / "dup")
        os.makedirs(d, exist_ok=True)
        with open(os.path.join(d, "part-00000.tfrecord"), "wb") as f:
            f.write(_raw_example_with_dup_keys())
        schema = stf.StructType([
            stf.StructField("x", stf.ArrayType(stf.LongType()), True)])
        got = stf.read_tfrecord(d, schema=schema, engine="gpu") \
            .to_arrow_table().column("x").to_pylist()
        assert got == [[7]]
