"""torch IterableDataset adapter tests (CPU engine; GPU covered in test_gpu)."""

import numpy as np
import torch

import spark_tfrecord_amd as stf
from spark_tfrecord_amd.torch_data import TFRecordIterableDataset


def _write_dataset(tmp, rows=5000, shards=4):
    out = str(tmp / "ds")
    rng = np.random.default_rng(0)
    data = {
        "uid": np.arange(rows, dtype=np.int64),
        "score": rng.random(rows).astype(np.float32),
        "tags": [[f"t{i % 5}"] * (i % 3) for i in range(rows)],
    }
    stf.write_tfrecord(data, out, engine="cpu", num_shards=shards)
    return out, data


class TestTorchDataset:
    def test_stream_all_rows(self, tmp_sandbox):
        out, data = _write_dataset(tmp_sandbox)
        ds = TFRecordIterableDataset(out, batch_rows=700, engine="cpu")
        got = []
        for b in ds:
            assert isinstance(b["uid"], torch.Tensor)
            assert b["uid"].numel() == int(b["_num_rows"])
            assert b["uid"].numel() <= 700
            got.append(b["uid"])
        uids = torch.cat(got).numpy()
        np.testing.assert_array_equal(np.sort(uids), data["uid"])

    def test_ragged_string_column(self, tmp_sandbox):
        out, data = _write_dataset(tmp_sandbox, rows=50, shards=1)
        ds = TFRecordIterableDataset(out, batch_rows=50, engine="cpu")
        b = next(iter(ds))
        off = b["tags_offsets"].numpy()
        voff = b["tags_value_offsets"].numpy()
        raw = bytes(b["tags"].numpy().tobytes())
        # row 4 has tags ["t4", "t4"] (i%3==1 -> 1 element? i=4: 4%3=1 -> 1)
        r = 4
        n_el = off[r + 1] - off[r]
        vals = [raw[voff[j]:voff[j + 1]].decode()
                for j in range(off[r], off[r + 1])]
        assert vals == data["tags"][r]
        assert n_el == len(data["tags"][r])

    def test_column_projection(self, tmp_sandbox):
        out, _ = _write_dataset(tmp_sandbox, rows=100, shards=2)
        ds = TFRecordIterableDataset(out, batch_rows=1000, engine="cpu",
                                     columns=["uid"])
        b = next(iter(ds))
        assert "uid" in b and "score" not in b

    def test_dataloader_workers_partition_files(self, tmp_sandbox):
        out, data = _write_dataset(tmp_sandbox, rows=4000, shards=8)
        ds = TFRecordIterableDataset(out, batch_rows=10_000, engine="cpu")
        dl = torch.utils.data.DataLoader(ds, batch_size=None, num_workers=2)
        total = sum(int(b["_num_rows"]) for b in dl)
        assert total == 4000

    def test_prefetch_zero_and_default_agree(self, tmp_sandbox):
        out, data = _write_dataset(tmp_sandbox, rows=2000, shards=5)
        import spark_tfrecord_amd.torch_data as td
        a = [int(b["_num_rows"]) for b in td.TFRecordIterableDataset(
            out, batch_rows=999, engine="cpu", prefetch=0)]
        b = [int(x["_num_rows"]) for x in td.TFRecordIterableDataset(
            out, batch_rows=999, engine="cpu", prefetch=2)]
        assert sum(a) == sum(b) == 2000
        assert a == b  # same shard/batch order

    def test_prefetch_propagates_errors(self, tmp_sandbox):
        import pytest

        out, _ = _write_dataset(tmp_sandbox, rows=100, shards=1)
        # corrupt the shard
        import os as _os
        f = [p for p in _os.listdir(out) if p.startswith("part-")][0]
        path = _os.path.join(out, f)
        blob = bytearray(open(path, "rb").read())
        blob[20] ^= 0xFF
        open(path, "wb").write(bytes(blob))
        ds = TFRecordIterableDataset(out, engine="cpu", prefetch=2,
                                     schema=None)
        with pytest.raises(RuntimeError):
            list(ds)
