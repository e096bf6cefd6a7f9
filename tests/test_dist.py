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
