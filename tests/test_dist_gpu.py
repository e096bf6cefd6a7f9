"""GPU-side distributed-layer tests (single-rank RCCL on the 1-GPU lease).

The driver's multi-GPU SCALE run is the only place 8 ranks exist; what CAN
be proven on one MI355X is that the RCCL ("nccl" on ROCm) code paths of
parallel/dist.py initialize, move DEVICE tensors through every collective
the engine uses (all-reduce MAX, all_gather_object, broadcast, all_to_all),
and agree with the gloo-tested semantics. World size 1 makes each
collective an identity, so results are exactly checkable."""

import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def nccl_group():
    import torch.distributed as dist

    if dist.is_initialized():  # pragma: no cover
        yield dist
        return
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", str(29810 + os.getpid() % 100))
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=0, world_size=1)
    yield dist
    dist.destroy_process_group()


class TestSingleRankRccl:
    def test_allreduce_max_lattice_codes(self, nccl_group):
        # the schema-inference merge: int64 lattice codes, MAX all-reduce
        dist = nccl_group
        vec = torch.tensor([0, 3, 7, 9], dtype=torch.int64, device="cuda")
        dist.all_reduce(vec, op=dist.ReduceOp.MAX)
        assert vec.cpu().tolist() == [0, 3, 7, 9]

    def test_broadcast_and_gather_object(self, nccl_group):
        dist = nccl_group
        t = torch.tensor([42], dtype=torch.int64, device="cuda")
        dist.broadcast(t, src=0)
        assert int(t.item()) == 42
        out = [None]
        dist.all_gather_object(out, {"a": 1})
        assert out == [{"a": 1}]

    def test_all_to_all_blobs_device_tensors(self, nccl_group):
        from spark_tfrecord_amd.parallel.dist import _all_to_all_blobs

        dev = torch.device("cuda", 0)
        blob = torch.arange(256, dtype=torch.int32).to(torch.uint8).to(dev)
        recv = _all_to_all_blobs([blob], dev)
        assert len(recv) == 1 and recv[0].is_cuda
        assert torch.equal(recv[0], blob)

    def test_all_to_all_blobs_bytes_and_empty(self, nccl_group):
        from spark_tfrecord_amd.parallel.dist import _all_to_all_blobs

        dev = torch.device("cuda", 0)
        recv = _all_to_all_blobs([b"\x01\x02\x03"], dev)
        assert bytes(recv[0].cpu().numpy().tobytes()) == b"\x01\x02\x03"
        recv = _all_to_all_blobs([b""], dev)
        assert recv[0].numel() == 0

    def test_infer_schema_distributed_gpu_kernel_path(self, nccl_group,
                                                      tmp_path):
        """world>1-style flow manually: the rank scans ITS slice with the
        device lattice kernel and the merged codes rebuild the schema."""
        import spark_tfrecord_amd as stf
        from spark_tfrecord_amd import _native
        from spark_tfrecord_amd.engine import gpu as gpu_engine
        from spark_tfrecord_amd.infer import schema_from_codes
        from spark_tfrecord_amd.io import paths as P

        out = str(tmp_path / "inf")
        stf.write_tfrecord({"a": [1.5, 2.5], "b": [[1, 2], [3]]}, out,
                           schema=stf.StructType([
                               stf.StructField("a", stf.FloatType(), True),
                               stf.StructField("b", stf.ArrayType(stf.LongType()), True)]))
        f = next(os.path.join(out, x) for x in sorted(os.listdir(out))
                 if x.startswith("part-"))
        data = np.frombuffer(P.decompress_file(f), np.uint8)
        off, lens = _native.scan_frames(data, False)
        dev_data = torch.as_tensor(np.ascontiguousarray(data)).cuda()
        codes = gpu_engine.infer_codes_device(
            dev_data, torch.as_tensor(np.ascontiguousarray(off)).cuda(),
            torch.as_tensor(np.ascontiguousarray(lens)).cuda(), "Example")
        # all-reduce the aligned code vector like infer_schema_distributed
        dist = nccl_group
        names = sorted(codes)
        vec = torch.tensor([codes[n] for n in names], dtype=torch.int64,
                           device="cuda")
        dist.all_reduce(vec, op=dist.ReduceOp.MAX)
        schema = schema_from_codes({n: int(v) for n, v in zip(names, vec)})
        assert schema["a"].dataType == stf.FloatType()
        assert schema["b"].dataType == stf.ArrayType(stf.LongType())

    def test_write_read_distributed_entrypoints(self, nccl_group, tmp_path):
        """The public distributed entry points run under an initialized nccl
        group (world 1 short-circuits to the local path — the point is that
        initialization state doesn't break them on a GPU rank)."""
        from spark_tfrecord_amd.parallel import (
            read_tfrecord_distributed,
            write_tfrecord_distributed,
        )

        out = str(tmp_path / "w1")
        write_tfrecord_distributed(
            {"x": np.arange(10, dtype=np.int64)}, out, mode="overwrite")
        df = read_tfrecord_distributed(out)
        assert sorted(r["x"] for r in df.collect()) == list(range(10))
