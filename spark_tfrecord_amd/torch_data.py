"""Training-pipeline integration: TFRecord shards as a torch IterableDataset.

The reference stops at Spark DataFrames; the MI355X-native equivalent of
"TFRecord -> accelerator" is streaming decoded shards straight into device
tensors (decode happens ON the training GPU — bytes cross PCIe once, columns
never touch host). Files are the sharding unit (matching the reference's
one-task-per-file model, DefaultSource.scala:26-29): they are split across
torch.distributed ranks and DataLoader workers round-robin.

    ds = TFRecordIterableDataset(path, batch_rows=8192)
    for batch in ds:           # dict: name -> values tensor,
        ...                    #       name+"_offsets" -> row offsets (ragged)
"""

from __future__ import annotations

import os
from typing import Dict, Iterator, List, Optional, Sequence

import numpy as np
import torch

from . import engine as engine_mod
from .infer import byte_array_schema
from .io import paths as P
from .schema import KIND_BYTES, StructType
from .columnar import RecordBatch

__all__ = ["TFRecordIterableDataset"]


def _as_torch(x, device):
    if isinstance(x, torch.Tensor):
        return x if device is None else x.to(device)
    t = torch.from_numpy(np.ascontiguousarray(x))
    return t if device is None else t.to(device)


def _slice_col(col, lo: int, hi: int, device):
    """Row-range [lo, hi) of one wire column as torch tensors."""
    row_off = _as_torch(col.row_off, None)
    v0 = int(row_off[lo])
    v1 = int(row_off[hi])
    out = {
        "offsets": _as_torch(col.row_off, device)[lo:hi + 1] - v0,
        "presence": _as_torch(col.presence, device)[lo:hi],
    }
    values = _as_torch(col.values, device)
    if col.kind == KIND_BYTES:
        elem_off = _as_torch(col.elem_off, None)
        b0 = int(elem_off[v0])
        b1 = int(elem_off[v1])
        out["values"] = values[b0:b1]
        out["value_offsets"] = _as_torch(col.elem_off, device)[v0:v1 + 1] - b0
    else:
        out["values"] = values[v0:v1]
    if col.is_seq and col.list_off is not None:
        list_off = _as_torch(col.list_off, None)
        l0 = int(list_off[lo])
        l1 = int(list_off[hi])
        out["list_offsets"] = _as_torch(col.list_off, device)[lo:hi + 1] - l0
        sub = _as_torch(col.sub_off, device)[l0:l1 + 1]
        out["sub_offsets"] = sub - sub[0]
    return out


class TFRecordIterableDataset(torch.utils.data.IterableDataset):
    """Iterates dicts of tensors over a TFRecord dataset.

    Each yielded item covers up to `batch_rows` rows of one shard:
      {col: values, col+"_offsets": per-row value offsets,
       col+"_presence": u8 mask, and for string/seq columns also
       col+"_value_offsets" / col+"_list_offsets" / col+"_sub_offsets"}.

    `device` "cuda": files decode on the GPU and tensors stay device-resident.
    Files are sharded across dist ranks (rank r takes files r, r+W, ...)
    and then across DataLoader workers the same way.
    """

    def __init__(self, path: str, schema: Optional[StructType] = None,
                 record_type: str = "Example", batch_rows: int = 65536,
                 columns: Optional[Sequence[str]] = None,
                 engine: str = "auto", verify_crc: bool = True,
                 shuffle_files: bool = False, seed: int = 0,
                 prefetch: int = 1):
        super().__init__()
        self.path = path
        self.record_type = record_type
        self.batch_rows = int(batch_rows)
        self.columns = list(columns) if columns else None
        self.engine = engine
        self.verify_crc = verify_crc
        self.shuffle_files = shuffle_files
        self.seed = seed
        self.prefetch = int(prefetch)  # shards decoded ahead (0 = sync)
        files = P.list_data_files(path)
        if not files:
            raise FileNotFoundError(f"No TFRecord files found under {path}")
        self.files: List[str] = files
        if schema is None:
            if record_type == "ByteArray":
                schema = byte_array_schema()
            else:
                from .io.reader import infer_schema_of_paths
                schema = infer_schema_of_paths(
                    files, record_type,
                    engine_mod.resolve_engine(engine)
                    if engine != "auto" else "cpu")
        self.schema = schema

    # -- sharding ---------------------------------------------------------
    def _my_files(self) -> List[str]:
        files = list(self.files)
        if self.shuffle_files:
            rng = np.random.default_rng(self.seed)
            rng.shuffle(files)
        try:
            import torch.distributed as dist
            if dist.is_available() and dist.is_initialized():
                files = files[dist.get_rank()::dist.get_world_size()]
        except Exception:
            pass
        info = torch.utils.data.get_worker_info()
        if info is not None:
            files = files[info.id::info.num_workers]
        return files

    def _decode_one(self, fpath: str, eng: str, sub_schema):
        if os.path.getsize(fpath) == 0:
            return None
        if eng == "gpu" and P.codec_from_path(fpath) is None:
            from .engine import gpu as gpu_engine
            batch = gpu_engine.read_file_to_batch(
                fpath, sub_schema, self.record_type, self.verify_crc)
            torch.cuda.current_stream().synchronize()  # hand off across threads
            return batch
        if eng == "gpu" and P.codec_from_path(fpath) == "gzip":
            # our gzip shards inflate ON the training GPU (segment table)
            from .engine import gpu as gpu_engine
            dev = gpu_engine.read_gzip_file_to_device(fpath)
            if dev is not None:
                if dev.numel() == 0:
                    return None
                off, lens = gpu_engine.scan_frames_device(dev)
                batch = gpu_engine.decode_device(
                    dev, off, lens, sub_schema, self.record_type,
                    self.verify_crc)
                torch.cuda.current_stream().synchronize()
                return batch
        from .engine import cpu as cpu_engine
        data = np.frombuffer(P.decompress_file(fpath), np.uint8)
        if data.size == 0:
            return None
        return cpu_engine.decode_buffer(data, sub_schema, self.record_type,
                                        self.verify_crc)

    def __iter__(self) -> Iterator[Dict[str, torch.Tensor]]:
        eng = engine_mod.resolve_engine(self.engine)
        fields = [f for f in self.schema.fields
                  if self.columns is None or f.name in self.columns]
        sub_schema = StructType(fields)
        files = self._my_files()
        if self.prefetch <= 0:
            for fpath in files:
                batch = self._decode_one(fpath, eng, sub_schema)
                if batch is not None:
                    yield from self._emit(batch, fields, None)
            return
        # background prefetch: a worker thread decodes the next shard(s) on
        # its own CUDA stream while the consumer drains the current one
        import queue
        import threading

        q: "queue.Queue" = queue.Queue(maxsize=self.prefetch)
        SENTINEL = object()

        def worker():
            try:
                stream = (torch.cuda.Stream()
                          if eng == "gpu" and torch.cuda.is_available() else None)
                for fpath in files:
                    if stream is not None:
                        with torch.cuda.stream(stream):
                            b = self._decode_one(fpath, eng, sub_schema)
                    else:
                        b = self._decode_one(fpath, eng, sub_schema)
                    if b is not None:
                        q.put(b)
                q.put(SENTINEL)
            except BaseException as e:  # noqa: BLE001 — surface in consumer
                q.put(e)

        t = threading.Thread(target=worker, daemon=True)
        t.start()
        while True:
            item = q.get()
            if item is SENTINEL:
                break
            if isinstance(item, BaseException):
                raise item
            yield from self._emit(item, fields, None)

    def _emit(self, batch: RecordBatch, fields, device):
        R = batch.num_rows
        for lo in range(0, R, self.batch_rows):
            hi = min(R, lo + self.batch_rows)
            out: Dict[str, torch.Tensor] = {}
            for f, col in zip(fields, batch.columns):
                parts = _slice_col(col, lo, hi, device)
                out[f.name] = parts["values"]
                out[f.name + "_offsets"] = parts["offsets"]
                out[f.name + "_presence"] = parts["presence"]
                for k in ("value_offsets", "list_offsets", "sub_offsets"):
                    if k in parts:
                        out[f"{f.name}_{k}"] = parts[k]
            out["_num_rows"] = torch.tensor(hi - lo)
            yield out
