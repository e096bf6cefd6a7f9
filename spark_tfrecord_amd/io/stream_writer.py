"""Streaming shard writer: incremental batches -> one TFRecord part file.

The reference's per-task writer is incremental — `OutputWriter.write(row)`
appends one framed record at a time (TFRecordOutputWriter.scala:26-38).
This is the batch-granular equivalent for producers that generate data in
chunks (training loops, ETL stages): each `write()` encodes its rows (on
the GPU when available) and appends the frames to the open shard; gzip
output streams through one compressobj with the library's full-flush
segment boundaries, so the finished file is identical in kind to
`write_tfrecord`'s.

    with ShardWriter(path + "/part-00000.tfrecord", schema) as w:
        for chunk in produce():
            w.write(chunk)
"""

from __future__ import annotations

import os
import zlib
from typing import Optional

from .. import engine as engine_mod
from ..arrow_interop import schema_from_arrow, table_to_batch
from ..schema import StructType, validate_schema_for_record_type
from . import paths as P
from .writer import normalize_input

__all__ = ["ShardWriter"]


class ShardWriter:
    def __init__(self, path: str, schema: Optional[StructType] = None,
                 record_type: str = "Example", codec: Optional[str] = None,
                 engine: str = "auto"):
        if record_type not in ("Example", "SequenceExample", "ByteArray"):
            raise ValueError(f"Unsupported recordType {record_type!r}")
        self.path = path
        self.schema = schema
        self.record_type = record_type
        self.codec = P.normalize_codec(codec)
        self._eng = engine_mod.resolve_engine(engine)
        self.rows_written = 0
        self._tmp = P.hidden_tmp_path(path)
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        self._f = open(self._tmp, "wb")
        self._gz = (zlib.compressobj(6, zlib.DEFLATED, 16 + 15)
                    if self.codec == "gzip" else
                    zlib.compressobj(6) if self.codec == "deflate" else None)
        self._closed = False

    # -- context manager --------------------------------------------------
    def __enter__(self):
        return self

    def __exit__(self, exc_type, exc, tb):
        if exc_type is None:
            self.close()
        else:
            self.abort()
        return False

    # -- writing -----------------------------------------------------------
    def write(self, data) -> int:
        """Append a chunk of rows; returns rows written so far."""
        if self._closed:
            raise RuntimeError("writer is closed")
        table = normalize_input(data, self.schema)
        if self.schema is None:
            self.schema = schema_from_arrow(table.schema)
        validate_schema_for_record_type(self.schema, self.record_type)
        batch = table_to_batch(table, self.schema)
        if self._eng == "gpu":
            from ..engine import gpu as gpu_engine

            raw = gpu_engine.encode_batch_from_cpu(batch, self.record_type)
        else:
            from ..engine import cpu as cpu_engine

            raw = cpu_engine.encode_batch(batch, self.record_type)
        if self._gz is not None:
            self._f.write(self._gz.compress(raw))
            self._f.write(self._gz.flush(zlib.Z_FULL_FLUSH))
        else:
            self._f.write(raw)
        self.rows_written += table.num_rows
        return self.rows_written

    def close(self):
        """Finish the stream and atomically publish the shard."""
        if self._closed:
            return
        if self._gz is not None:
            self._f.write(self._gz.flush())
        self._f.flush()
        os.fsync(self._f.fileno())
        self._f.close()
        os.replace(self._tmp, self.path)
        self._closed = True

    def abort(self):
        """Drop the partial shard (nothing becomes visible)."""
        if self._closed:
            return
        self._f.close()
        try:
            os.unlink(self._tmp)
        except FileNotFoundError:
            pass
        self._closed = True
