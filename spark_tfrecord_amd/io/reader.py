"""Read path: TFRecord files -> DataFrame.

Mirrors the reference's read pipeline: file discovery
(whole files, never split — gzip streams can't be, DefaultSource.scala:26-29),
schema inference from the first non-empty file when no schema is given,
per-file decode, and partition-column discovery from `col=value/` directory
names (the part Spark's planner did above the reference library).
"""

from __future__ import annotations

import os
from typing import List, Optional

import numpy as np
import pyarrow as pa

from .. import engine as engine_mod
from ..arrow_interop import batch_to_table, schema_to_arrow
from ..engine import cpu as cpu_engine
from ..infer import (
    byte_array_schema,
    infer_codes_from_buffer,
    merge_code_maps,
    schema_from_codes,
)
from ..schema import (LongType, StringType, StructField, StructType,
                      validate_schema_for_record_type)
from ..utils import IOMetrics, StageTimer
from .. import _native
from . import paths as P

__all__ = ["read_tfrecord", "infer_schema_of_paths"]


def _load_file(path: str) -> np.ndarray:
    return np.frombuffer(P.decompress_file(path), np.uint8)


def infer_schema_of_paths(files: List[str], record_type: str,
                          engine: str = "cpu") -> StructType:
    """Schema from the FIRST non-empty file (DefaultSource.scala:36-38
    collectFirst), scanned fully — on the GPU (hash-table lattice kernel)
    when engine='gpu' and the file is uncompressed."""
    if record_type == "ByteArray":
        return byte_array_schema()
    for f in files:
        if engine == "gpu" and P.codec_from_path(f) in (None, "gzip"):
            import os as _os

            if _os.path.getsize(f) == 0:
                continue
            from ..engine import gpu as gpu_engine

            if P.codec_from_path(f) == "gzip":
                data = gpu_engine.read_gzip_file_to_device(f)
                if data is None:  # foreign gzip: host inflate + lattice
                    data_np = _load_file(f)
                    if data_np.size == 0:
                        continue
                    off_h, lens_h = _native.scan_frames(data_np, False)
                    if len(off_h) == 0:
                        continue
                    codes = infer_codes_from_buffer(data_np, off_h, lens_h,
                                                    record_type)
                    if codes:
                        return schema_from_codes(codes)
                    return StructType([])
                if data.numel() == 0:
                    continue
            else:
                data = gpu_engine.read_file_to_device(f)
            off, lens = gpu_engine.scan_frames_device(data)
            if off.numel() == 0:
                continue
            codes = gpu_engine.infer_codes_device(data, off, lens, record_type)
        else:
            data = _load_file(f)
            if data.size == 0:
                continue
            off, lens = _native.scan_frames(data, False)
            if len(off) == 0:
                continue
            codes = infer_codes_from_buffer(data, off, lens, record_type)
        if codes:
            return schema_from_codes(codes)
        return StructType([])
    raise ValueError("Could not infer schema: no non-empty TFRecord files found")


def _partition_schema(files: List[str], base_dir: str) -> List[str]:
    cols: List[str] = []
    for f in files:
        for k in P.partition_values_of(f, base_dir):
            if k not in cols:
                cols.append(k)
    return cols


def _partition_col_array(values: List[str], n: int):
    """Spark-style partition value type inference: int64 if all parse, else
    string."""
    try:
        return pa.array([int(v) for v in values], type=pa.int64())
    except (ValueError, TypeError):
        return pa.array(
            [None if v == "__HIVE_DEFAULT_PARTITION__" else v for v in values],
            type=pa.large_utf8())


def count_tfrecord(path: str, engine: str = "auto") -> int:
    """Total record count without decoding any payloads (frame scan only —
    the GPU path discovers and chains frame boundaries in parallel)."""
    files = P.list_data_files(path)
    if not files:
        raise FileNotFoundError(f"No TFRecord files found under {path}")
    eng = engine_mod.resolve_engine(engine)
    total = 0
    for f in files:
        if eng == "gpu" and P.codec_from_path(f) in (None, "gzip"):
            if os.path.getsize(f) == 0:
                continue
            from ..engine import gpu as gpu_engine

            if P.codec_from_path(f) == "gzip":
                data = gpu_engine.read_gzip_file_to_device(f)
                if data is None:  # foreign gzip: host count below
                    data_np = _load_file(f)
                    if data_np.size:
                        off_h, _ = _native.scan_frames(data_np, False)
                        total += len(off_h)
                    continue
                if data.numel() == 0:
                    continue
            else:
                data = gpu_engine.read_file_to_device(f)
            off, _ = gpu_engine.scan_frames_device(data)
            total += int(off.numel())
        else:
            data = _load_file(f)
            if data.size == 0:
                continue
            off, _ = _native.scan_frames(data, False)
            total += len(off)
    return total


def read_tfrecord(path: str, schema: Optional[StructType] = None,
                  record_type: str = "Example", engine: str = "auto",
                  verify_crc: bool = True, base_dir: Optional[str] = None,
                  columns: Optional[List[str]] = None):
    """Read a TFRecord dataset as a DataFrame.

    `columns` projects the read: only the named fields are scanned and
    extracted (the schema-driven kernels skip other features entirely, like
    the reference's deserializer ignores unknown ones)."""
    from ..api import DataFrame

    if record_type not in ("Example", "SequenceExample", "ByteArray"):
        raise ValueError(f"Unsupported recordType {record_type!r}")
    files = P.list_data_files(path)
    if not files:
        raise FileNotFoundError(f"No TFRecord files found under {path}")
    if base_dir is None:
        if isinstance(path, (list, tuple)):
            base_dir = (os.path.commonpath([os.path.dirname(f) for f in files])
                        if files else "")
        else:
            base_dir = path if os.path.isdir(path) else os.path.dirname(path)
    part_cols = _partition_schema(files, base_dir) if base_dir else []
    eng = engine_mod.resolve_engine(engine)

    metrics = IOMetrics("read")

    # Files needing host bytes (compressed, or CPU engine) are loaded and
    # inflated by a thread pool — gzip/zlib release the GIL, so multi-file
    # reads decompress in parallel (gzip itself is sequential PER file,
    # matching the reference's isSplitable=false model). A sliding window
    # bounds resident decompressed bytes.
    from concurrent.futures import ThreadPoolExecutor

    def _needs_host_bytes(fpath: str) -> bool:
        if eng != "gpu":
            return True
        codec = P.codec_from_path(fpath)
        if codec is None:
            return False
        if codec == "gzip":
            # our gzip files carry a segment table: the device inflater
            # decompresses them in HBM (foreign gzip stays on host zlib)
            from ..engine import gpu as gpu_engine
            return gpu_engine.gz_device_meta(fpath) is None
        return True

    # Schema-less GPU reads DEFER inference into the first group pipeline:
    # the file image is scanned once in HBM, the lattice kernel runs over
    # the first non-empty file's frames, and the decode reuses the image —
    # one pass instead of the reference's extra inference job.
    deferred_infer = (schema is None and eng == "gpu" and bool(files)
                      and not _needs_host_bytes(files[0]))
    if schema is None and not deferred_infer:
        with StageTimer(metrics, "infer_schema"):
            schema = (byte_array_schema() if record_type == "ByteArray"
                      else infer_schema_of_paths(files, record_type, eng))

    def _resolve(schema):
        nonlocal part_cols
        if columns is not None:
            missing = [c for c in columns
                       if c not in [f.name for f in schema.fields]
                       and c not in part_cols]
            if missing:
                raise KeyError(f"columns not in schema: {missing}")
            schema = StructType(
                [f for f in schema.fields if f.name in set(columns)])
            part_cols = [c for c in part_cols if c in set(columns)]
        data_schema = StructType(
            [f for f in schema.fields if f.name not in part_cols])
        # an Example cannot carry 2-D ragged fields: reject like the
        # reference's deserializer would (TFRecordDeserializer.scala:148-175)
        validate_schema_for_record_type(data_schema, record_type)
        return data_schema

    data_schema = None if deferred_infer else _resolve(schema)

    workers = min(32, (os.cpu_count() or 8))
    window = 2 * workers
    futures: dict = {}
    _pool = P.shared_pool  # persistent process-wide pool (no spin-up)

    def _host_task(fpath: str):
        data = _load_file(fpath)
        if data.size == 0 or eng == "gpu":
            return data  # GPU engine decodes on the device stream
        # the native codec releases the GIL: files decode concurrently
        return cpu_engine.decode_buffer(data, data_schema, record_type,
                                        verify_crc=verify_crc)

    # tiny jobs (<= 2 small files) skip the pool: submit/wait latency
    # dominates sub-ms decodes (the 1k-row plumbing config round trip)
    _inline_small = (len(files) <= 2 and
                     all(os.path.getsize(f) < (256 << 10) for f in files))

    def _blob(i: int):
        f = futures.pop(i, None)
        if f is not None:
            return f.result()
        if _inline_small:
            return _host_task(files[i])
        return _pool().submit(_host_task, files[i]).result()

    # GPU path: consecutive uncompressed files are decoded as ONE pipeline
    # (images concatenated in HBM — frames are concatenable — scanned and
    # decoded in a single pass, rows split per file afterwards). Compressed
    # files and the CPU engine go file-by-file through host bytes.
    _GROUP_BYTES = 4 << 30

    def _append_with_parts(t, fpaths, row_counts):
        pvs = [P.partition_values_of(f, base_dir) for f in fpaths]
        for c in part_cols:
            vals = np.repeat([pv.get(c) for pv in pvs], row_counts).tolist()
            t = t.append_column(c, _partition_col_array(vals, t.num_rows))
        tables.append(t)

    tables = []
    try:
        i = 0
        while i < len(files):
            fpath = files[i]
            if eng == "gpu" and not _needs_host_bytes(fpath):
                from ..engine import gpu as gpu_engine

                group = []
                gbytes = 0
                while (i < len(files) and not _needs_host_bytes(files[i])
                       and gbytes < _GROUP_BYTES):
                    group.append(files[i])
                    sz = os.path.getsize(files[i])
                    gbytes += sz
                    metrics.add(files=1, nbytes=sz)
                    i += 1
                if data_schema is None:
                    try:
                        batch, row_counts = gpu_engine.read_files_to_batch(
                            group, None, record_type, verify_crc=verify_crc)
                        data_schema = _resolve(batch.schema)
                    except ValueError:
                        # this group was entirely empty: resolve the schema
                        # from the full file list, then decode normally
                        data_schema = _resolve(
                            byte_array_schema() if record_type == "ByteArray"
                            else infer_schema_of_paths(files, record_type, eng))
                        batch, row_counts = gpu_engine.read_files_to_batch(
                            group, data_schema, record_type,
                            verify_crc=verify_crc)
                else:
                    batch, row_counts = gpu_engine.read_files_to_batch(
                        group, data_schema, record_type, verify_crc=verify_crc)
                if batch.num_rows == 0 and not part_cols:
                    continue
                t = batch_to_table(gpu_engine.batch_to_host(batch))
                if [f.name for f in data_schema.fields] != t.column_names:
                    # deferred inference + column projection: decode carried
                    # the full inferred schema; project here
                    t = t.select([f.name for f in data_schema.fields])
                _append_with_parts(t, group, row_counts)
                continue
            for j in range(i, min(i + window, len(files))):
                if (not _inline_small and _needs_host_bytes(files[j])
                        and j not in futures):
                    futures[j] = _pool().submit(_host_task, files[j])
            metrics.add(files=1, nbytes=os.path.getsize(fpath))
            got = _blob(i)
            i += 1
            if eng == "gpu":
                if got.size == 0:
                    continue
                from ..engine import gpu as gpu_engine
                batch = gpu_engine.decode_buffer_to_cpu(
                    got, data_schema, record_type, verify_crc=verify_crc)
            else:
                if isinstance(got, np.ndarray):  # empty file sentinel
                    continue
                batch = got
            t = batch_to_table(batch)
            _append_with_parts(t, [fpath], [t.num_rows])
    finally:
        for f in futures.values():  # abandon prefetches from a failed read
            f.cancel()
    if not tables:
        full = StructType(list(data_schema.fields) +
                          [StructField(c, StringType(), True) for c in part_cols])
        empty = pa.table({f.name: [] for f in full.fields},
                         schema=schema_to_arrow(full))
        return DataFrame(empty, full)

    # Partition column types must agree across files: rebuild as common type
    table = pa.concat_tables(tables, promote_options="permissive")
    full_schema = StructType(list(data_schema.fields))
    for c in part_cols:
        at = table.schema.field(c).type
        full_schema.add(c, LongType() if pa.types.is_integer(at) else StringType())
    metrics.add(rows=table.num_rows)
    metrics.finish()
    return DataFrame(table, full_schema)
