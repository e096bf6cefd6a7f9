"""Streaming shard writer: incremental batches -> one TFRecord part file.

The reference's per-task writer is incremental — `OutputWriter.write(row)`
appends one framed record at a time (TFRecordOutputWriter.scala:26-38).
This is the batch-granular equivalent for producers that generate data in
chunks (training loops, ETL stages): each `write()` encodes its rows (on
the GPU when available) and appends the frames to the open shard.

gzip shards stream through ONE raw-deflate stream with the library's
full-flush 32 KiB segment boundaries, and the header reserves an FEXTRA
region that is backpatched with the 'TS' segment table on close — the
finished file is byte-compatible with `write_tfrecord`'s gzip output, so
reads inflate it ON DEVICE (one segment per lane/half-wave). If a shard
outgrows the reserved table (`segment_table_capacity`), the table is
dropped (padding only) and reads fall back to the full-flush marker scan,
exactly like a foreign gzip file.

    with ShardWriter(path + "/part-00000.tfrecord", schema) as w:
        for chunk in produce():
            w.write(chunk)
"""

from __future__ import annotations

import os
import struct
import zlib
from typing import List, Optional, Tuple

from .. import engine as engine_mod
from ..arrow_interop import schema_from_arrow, table_to_batch
from ..schema import StructType, validate_schema_for_record_type
from . import paths as P
from .writer import normalize_input

__all__ = ["ShardWriter"]


class ShardWriter:
    def __init__(self, path: str, schema: Optional[StructType] = None,
                 record_type: str = "Example", codec: Optional[str] = None,
                 engine: str = "auto", segment_table_capacity: int = 2048):
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
        self._gz = None
        self._zl = None
        if self.codec == "gzip":
            # raw-deflate stream + hand-rolled gzip framing so the header
            # can reserve the FEXTRA segment-table region (backpatched on
            # close; zlib's own gzip wrapper would own the header)
            self._gz = zlib.compressobj(6, zlib.DEFLATED, -15)
            self._cap = max(1, min(int(segment_table_capacity), P._GZ_MAX_SEGS))
            self._xlen = 4 + (4 + 8 * self._cap)  # full TS subfield
            if self._xlen > 0xFFFF:
                raise ValueError("segment_table_capacity too large for FEXTRA")
            self._f.write(b"\x1f\x8b\x08\x04" + b"\x00\x00\x00\x00" +
                          b"\x00\xff" + struct.pack("<H", self._xlen) +
                          b"\x00" * self._xlen)
            self._segs: List[Tuple[int, int]] = []  # (comp_len, uncomp_len)
            self._crc = 0
            self._isize = 0
            self._pending = 0   # uncompressed bytes in the OPEN segment
            self._seg_comp = 0  # compressed bytes already written for it
        elif self.codec == "deflate":
            self._zl = zlib.compressobj(6)
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

    # -- gzip segment stream ----------------------------------------------
    def _gz_append(self, raw: bytes, final: bool = False):
        """Feed `raw` through the deflate stream in 32 KiB full-flush
        segments (the same boundaries `compress_bytes` emits)."""
        seg = P._GZ_SEGMENT
        self._crc = zlib.crc32(raw, self._crc)
        self._isize = (self._isize + len(raw)) % (1 << 32)
        pos, n = 0, len(raw)
        while True:
            take = min(seg - self._pending, n - pos)
            if take > 0:
                body = self._gz.compress(raw[pos:pos + take])
                self._f.write(body)
                self._seg_comp += len(body)
                pos += take
                self._pending += take
            closing = final and pos >= n
            if closing or self._pending == seg:
                body = (self._gz.flush() if closing
                        else self._gz.flush(zlib.Z_FULL_FLUSH))
                self._f.write(body)
                self._seg_comp += len(body)
                self._segs.append((self._seg_comp, self._pending))
                self._seg_comp = 0
                self._pending = 0
                if closing:
                    return
            if pos >= n:
                return

    def _gz_finish(self):
        self._gz_append(b"", final=True)
        self._f.write(struct.pack("<II", self._crc & 0xFFFFFFFF, self._isize))
        # backpatch the reserved FEXTRA region: the real TS table when it
        # fits, otherwise padding only (readers then use the marker scan)
        segs = self._segs if len(self._segs) <= self._cap else []
        ts_payload = struct.pack("<BBH", 1, 0, len(segs)) + b"".join(
            struct.pack("<II", c, u) for c, u in segs)
        extra = b"TS" + struct.pack("<H", len(ts_payload)) + ts_payload
        pad = self._xlen - len(extra)
        if pad:
            # private 'ZP' subfield consumes the rest; readers skip it
            extra += b"ZP" + struct.pack("<H", pad - 4) + b"\x00" * (pad - 4)
        assert len(extra) == self._xlen
        self._f.seek(12)
        self._f.write(extra)
        self._f.seek(0, os.SEEK_END)

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
        raw = bytes(raw)
        if self._gz is not None:
            self._gz_append(raw)
        elif self._zl is not None:
            self._f.write(self._zl.compress(raw))
            self._f.write(self._zl.flush(zlib.Z_FULL_FLUSH))
        else:
            self._f.write(raw)
        self.rows_written += table.num_rows
        return self.rows_written

    def close(self):
        """Finish the stream and atomically publish the shard."""
        if self._closed:
            return
        if self._gz is not None:
            self._gz_finish()
        elif self._zl is not None:
            self._f.write(self._zl.flush())
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
