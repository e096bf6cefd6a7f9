"""Multi-process distributed tests (gloo backend, world_size=2, CPU).

The reference tests distributed behavior on Spark local mode without a
cluster (SharedSparkSessionSuite); the analog here is torch.distributed over
gloo with two processes on 127.0.0.1 — the same code paths the GPU runs with
RCCL (backend selection is the only difference)."""

import os

import numpy as np
import pytest
import torch.multiprocessing as mp

WORLD = 2


def _run(fn, tmp, extra=None, world=WORLD):
    ctx = mp.get_context("spawn")
    port = 29600 + (os.getpid() % 500)
    procs = [ctx.Process(target=_entry,
                         args=(fn.__name__, r, port, str(tmp), extra, world))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(180)
    codes = [p.exitcode for p in procs]
    for p in procs:
        if p.is_alive():
            p.terminate()
    assert codes == [0] * world, f"worker exit codes: {codes}"


def _entry(fn_name, rank, port, tmp, extra, world=WORLD):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    os.environ["TFREC_FORCE_CPU"] = "1"
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        globals()[fn_name](rank, tmp, extra)
        dist.barrier()
    finally:
        dist.destroy_process_group()


# -- worker bodies (module-level so spawn can import them) -------------------

def _w_plain_write(rank, tmp, extra):
    import spark_tfrecord_amd as stf
    from spark_tfrecord_amd.parallel import (
        read_tfrecord_distributed,
        write_tfrecord_distributed,
    )

    out = os.path.join(tmp, "plain")
    data = {"id": np.arange(rank * 10, rank * 10 + 10, dtype=np.int64),
            "v": [f"r{rank}-{i}" for i in range(10)]}
    write_tfrecord_distributed(data, out, mode="overwrite")
    # every rank reads its shard; union must be all 20 rows
    df = read_tfrecord_distributed(out)
    import torch.distributed as dist

    local_ids = sorted(r["id"] for r in df.collect())
    gathered = [None, None]
    dist.all_gather_object(gathered, local_ids)
    all_ids = sorted(x for g in gathered for x in g)
    assert all_ids == list(range(20)), all_ids
    assert os.path.exists(os.path.join(out, "_SUCCESS"))
    parts = [f for f in os.listdir(out) if f.startswith("part-")]
    assert len(parts) == WORLD


def _w_partitioned(rank, tmp, extra):
    from spark_tfrecord_amd.parallel import write_tfrecord_distributed

    out = os.path.join(tmp, "parts")
    # both ranks hold rows of BOTH partition values -> all-to-all must merge
    data = {"date": ["d1", "d2", "d1", "d2"],
            "x": np.array([0, 1, 2, 3], np.int64) + 10 * rank}
    write_tfrecord_distributed(data, out, partition_by=["date"], mode="overwrite")
    import torch.distributed as dist

    dist.barrier()
    if rank == 0:
        import spark_tfrecord_amd as stf

        assert sorted(d for d in os.listdir(out) if d.startswith("date=")) == \
            ["date=d1", "date=d2"]
        # each partition dir was written by exactly ONE rank
        for d in ["date=d1", "date=d2"]:
            parts = os.listdir(os.path.join(out, d))
            assert len(parts) == 1, parts
        df = stf.read_tfrecord(out).sort("x")
        rows = df.collect()
        assert [r["x"] for r in rows] == [0, 1, 2, 3, 10, 11, 12, 13]
        assert {r["date"] for r in rows} == {"d1", "d2"}


def _w_infer(rank, tmp, extra):
    import spark_tfrecord_amd as stf
    from spark_tfrecord_amd.parallel import infer_schema_distributed

    # rank 0 writes one file whose records REQUIRE merging across ranks'
    # slices: record 0 (rank 0's slice) is long, record 1 (rank 1) is float
    path = os.path.join(tmp, "inf")
    if rank == 0:
        stf.write_tfrecord(
            {"a": [1.0, 2.0], "b": [[1, 2], [3, 4]]}, path, mode="overwrite",
            schema=stf.StructType([
                stf.StructField("a", stf.FloatType(), True),
                stf.StructField("b", stf.ArrayType(stf.LongType()), True)]))
    import torch.distributed as dist

    dist.barrier()
    files = sorted(
        os.path.join(path, f) for f in os.listdir(path) if f.startswith("part-"))
    schema = infer_schema_distributed(files, "Example")
    assert schema["a"].dataType == stf.FloatType()
    assert schema["b"].dataType == stf.ArrayType(stf.LongType())


def _w_save_mode_ignore(rank, tmp, extra):
    from spark_tfrecord_amd.parallel import write_tfrecord_distributed

    out = os.path.join(tmp, "ig")
    data = {"x": np.array([rank], np.int64)}
    write_tfrecord_distributed(data, out, mode="overwrite")
    import torch.distributed as dist

    dist.barrier()
    before = sorted(os.listdir(out))
    dist.barrier()
    write_tfrecord_distributed({"x": np.array([99], np.int64)}, out, mode="ignore")
    dist.barrier()
    assert sorted(os.listdir(out)) == before


def _w_pipelined_write(rank, tmp, extra):
    """Each rank writes its own part file through the PIPELINED single-shard
    path (chunk appends overlapped with encode); a rank-sharded read must
    see every row exactly once."""
    import spark_tfrecord_amd as stf
    from spark_tfrecord_amd.io import writer as W
    from spark_tfrecord_amd.parallel import write_tfrecord_distributed

    W._PIPE_CHUNK_ROWS = 400  # force many chunks per rank
    out = os.path.join(tmp, "pipew")
    rows = 3_000
    data = {"x": np.arange(rank * rows, (rank + 1) * rows, dtype=np.int64)}
    write_tfrecord_distributed(data, out, mode="overwrite")
    import torch.distributed as dist

    dist.barrier()
    if rank == 0:
        df = stf.read_tfrecord(out, engine="cpu")
        got = sorted(r["x"] for r in df.collect())
        import torch.distributed as d2
        world = d2.get_world_size()
        assert got == list(range(world * rows))


# -- pytest entry points -----------------------------------------------------

@pytest.mark.timeout(240)
def test_distributed_plain_write_read(tmp_path):
    _run(_w_plain_write, tmp_path)


@pytest.mark.timeout(240)
def test_distributed_partitioned_write(tmp_path):
    _run(_w_partitioned, tmp_path)


@pytest.mark.timeout(240)
def test_distributed_schema_inference(tmp_path):
    _run(_w_infer, tmp_path)


@pytest.mark.timeout(240)
def test_distributed_save_mode_ignore(tmp_path):
    _run(_w_save_mode_ignore, tmp_path)


def _w_blob_exchange(rank, tmp, extra):
    """Every branch of _all_to_all_blobs on gloo: bytes in, tensors in,
    zero-length blobs, wildly uneven sizes. extra = per-blob scale factor."""
    import torch
    import torch.distributed as dist

    from spark_tfrecord_amd.parallel.dist import _all_to_all_blobs

    world = dist.get_world_size()
    scale = int(extra or 1)

    def payload(src, dst):
        if (src + dst) % 3 == 0:
            return b""  # empty-blob branch
        n = ((src * 7 + dst * 13) % 11) * 1024 * scale + src + dst
        return bytes([((src * 251) ^ (dst * 17) ^ (i & 0xFF)) & 0xFF
                      for i in range(min(n, 64))]) * max(1, n // 64)

    send = [payload(rank, d) for d in range(world)]
    if rank % 2 == 1:  # tensor-input branch
        send = [torch.frombuffer(bytearray(b), dtype=torch.uint8)
                if len(b) else torch.zeros(0, dtype=torch.uint8)
                for b in send]
    recv = _all_to_all_blobs(send, torch.device("cpu"))
    assert len(recv) == world
    for src in range(world):
        got = recv[src]
        if hasattr(got, "numpy"):
            got = bytes(got.cpu().numpy().tobytes())
        want = payload(src, rank)
        want_n = len(want) if isinstance(want, (bytes, bytearray)) else want.numel()
        assert len(got) == want_n, (rank, src, len(got), want_n)
        assert got == (want if isinstance(want, bytes) else bytes(want)), \
            (rank, src)


def _w_partitioned_uneven(rank, tmp, extra):
    """Skew: rank 0 holds ALL rows of partition 'hot'; ranks >= world//2
    hold no rows at all (zero-row ranks must still participate in every
    collective); 3 partitions < world so most ranks own no partition."""
    import torch.distributed as dist

    from spark_tfrecord_amd.parallel import write_tfrecord_distributed

    import pyarrow as pa

    world = dist.get_world_size()
    out = os.path.join(tmp, "uneven")
    if rank == 0:
        data = {"p": ["hot"] * 50 + ["a", "b"],
                "x": np.arange(52, dtype=np.int64)}
    elif rank < world // 2:
        data = {"p": ["a", "b"], "x": np.array([100 + rank, 200 + rank], np.int64)}
    else:
        data = pa.table({"p": pa.array([], type=pa.large_utf8()),
                         "x": pa.array([], type=pa.int64())})
    write_tfrecord_distributed(data, out, partition_by=["p"], mode="overwrite")
    dist.barrier()
    if rank == 0:
        import spark_tfrecord_amd as stf

        dirs = sorted(d for d in os.listdir(out) if d.startswith("p="))
        assert dirs == ["p=a", "p=b", "p=hot"], dirs
        df = stf.read_tfrecord(out)
        n_expected = 52 + 2 * (world // 2 - 1)
        assert df.count() == n_expected, (df.count(), n_expected)
        rows = df.collect()
        hot = sorted(r["x"] for r in rows if r["p"] == "hot")
        assert hot == list(range(50)), hot


def _w_infer_sparse(rank, tmp, extra):
    """Schema inference with fewer records than ranks: most ranks scan an
    EMPTY slice (codes = {}) and must still agree on the merged schema."""
    import torch.distributed as dist

    import spark_tfrecord_amd as stf
    from spark_tfrecord_amd.parallel import infer_schema_distributed

    path = os.path.join(tmp, "sparse")
    if rank == 0:
        # three records with DIFFERENT inferred types for 'v' in ONE file
        # (frames concatenate): long scalar, float scalar, long array
        os.makedirs(path, exist_ok=True)
        blobs = []
        for i, (d, dt) in enumerate([
                ({"v": [1]}, stf.LongType()),
                ({"v": [2.5]}, stf.FloatType()),
                ({"v": [[1, 2, 3]]}, stf.ArrayType(stf.LongType()))]):
            td = os.path.join(tmp, f"sparse_src{i}")
            stf.write_tfrecord(d, td, mode="overwrite", schema=stf.StructType(
                [stf.StructField("v", dt, True)]))
            part = next(f for f in os.listdir(td) if f.startswith("part-"))
            with open(os.path.join(td, part), "rb") as f:
                blobs.append(f.read())
        with open(os.path.join(path, "part-00000-m.tfrecord"), "wb") as f:
            f.write(b"".join(blobs))
    dist.barrier()
    files = [os.path.join(path, "part-00000-m.tfrecord")]
    schema = infer_schema_distributed(files, "Example")
    # precedence-max merge (TensorFlowInferSchema.scala:194-228):
    # Long < Float < Arr[Long] -> Arr[Long]
    assert schema["v"].dataType == stf.ArrayType(stf.LongType()), schema


def _w_infer_empty_dataset(rank, tmp, extra):
    """All files empty: every rank must raise the same ValueError."""
    import pytest
    import torch.distributed as dist

    from spark_tfrecord_amd.parallel import infer_schema_distributed

    path = os.path.join(tmp, "empty")
    if rank == 0:
        os.makedirs(path, exist_ok=True)
        open(os.path.join(path, "part-00000-x.tfrecord"), "wb").close()
    dist.barrier()
    files = [os.path.join(path, "part-00000-x.tfrecord")]
    with pytest.raises(ValueError, match="no non-empty"):
        infer_schema_distributed(files, "Example")


def _w_infer_bytearray(rank, tmp, extra):
    """ByteArray short-circuit: fixed schema, no file IO, no collectives."""
    import spark_tfrecord_amd as stf
    from spark_tfrecord_amd.parallel import infer_schema_distributed

    schema = infer_schema_distributed([], "ByteArray")
    assert [f.name for f in schema.fields] == ["byteArray"]
    assert isinstance(schema.fields[0].dataType, stf.BinaryType)


def _w_read_fewer_files_than_ranks(rank, tmp, extra):
    """Round-robin sharding with 2 files over a larger world: ranks beyond
    the file count get a valid EMPTY DataFrame with the right schema."""
    import torch.distributed as dist

    import spark_tfrecord_amd as stf
    from spark_tfrecord_amd.parallel import read_tfrecord_distributed

    world = dist.get_world_size()
    out = os.path.join(tmp, "few")
    if rank == 0:
        stf.write_tfrecord({"x": np.arange(6, dtype=np.int64)}, out,
                           mode="overwrite", num_shards=2)
    dist.barrier()
    df = read_tfrecord_distributed(out)
    local = sorted(r["x"] for r in df.collect())
    gathered = [None] * world
    dist.all_gather_object(gathered, local)
    assert sorted(x for g in gathered for x in g) == list(range(6))
    if rank >= 2:
        assert local == []
        assert [f.name for f in df.schema.fields] == ["x"]


def _w_partitioned_any_world(rank, tmp, extra):
    import torch.distributed as dist

    from spark_tfrecord_amd.parallel import write_tfrecord_distributed

    world = dist.get_world_size()
    out = os.path.join(tmp, "parts_any")
    data = {"date": [f"d{i % 3}" for i in range(6)],
            "x": np.arange(6, dtype=np.int64) + 100 * rank}
    write_tfrecord_distributed(data, out, partition_by=["date"],
                               mode="overwrite")
    dist.barrier()
    if rank == 0:
        import spark_tfrecord_amd as stf

        assert sorted(d for d in os.listdir(out) if d.startswith("date=")) == \
            ["date=d0", "date=d1", "date=d2"]
        for d in ["date=d0", "date=d1", "date=d2"]:
            # exactly one owner rank wrote each partition directory
            assert len(os.listdir(os.path.join(out, d))) == 1
        df = stf.read_tfrecord(out)
        assert df.count() == 6 * world
        xs = sorted(r["x"] for r in df.collect())
        want = sorted(i + 100 * r for r in range(world) for i in range(6))
        assert xs == want


def test_distributed_partitioned_write_world4(tmp_path):
    _run(_w_partitioned_any_world, tmp_path, world=4)


# -- world=8 hardening (the dp8 shapes the driver's SCALE run exercises) -----

@pytest.mark.timeout(420)
def test_blob_exchange_world8(tmp_path):
    _run(_w_blob_exchange, tmp_path, extra=1, world=8)


@pytest.mark.timeout(600)
@pytest.mark.skipif(os.environ.get("TFREC_BIG_DIST") != "1",
                    reason="set TFREC_BIG_DIST=1 for the GB-scale exchange")
def test_blob_exchange_world8_large(tmp_path):
    # ~11 MB x 8 peers x 8 ranks ~ 0.7 GB total in flight over gloo
    _run(_w_blob_exchange, tmp_path, extra=1024, world=8)


@pytest.mark.timeout(420)
def test_distributed_partitioned_uneven_world8(tmp_path):
    _run(_w_partitioned_uneven, tmp_path, world=8)


@pytest.mark.timeout(420)
def test_distributed_infer_sparse_world8(tmp_path):
    _run(_w_infer_sparse, tmp_path, world=8)


@pytest.mark.timeout(240)
def test_distributed_infer_empty_dataset(tmp_path):
    _run(_w_infer_empty_dataset, tmp_path)


@pytest.mark.timeout(240)
def test_distributed_infer_bytearray(tmp_path):
    _run(_w_infer_bytearray, tmp_path)


@pytest.mark.timeout(420)
def test_distributed_read_fewer_files_than_ranks(tmp_path):
    _run(_w_read_fewer_files_than_ranks, tmp_path, world=5)


@pytest.mark.timeout(240)
def test_distributed_pipelined_write(tmp_path):
    _run(_w_pipelined_write, tmp_path)
