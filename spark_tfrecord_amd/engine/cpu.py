"""CPU engine: host-codec encode/decode of whole file buffers.

Serves the no-GPU plumbing path (BASELINE.json config 1) and acts as the
golden reference for the GPU pipeline's numerics tests."""

from __future__ import annotations

from typing import Optional

import numpy as np

from .. import _native
from ..columnar import (
    RecordBatch,
    WireColumn,
    native_dict_to_wire,
    schema_blob,
    wire_fields,
    wire_to_native_dict,
)
from ..schema import StructType

FMT = {"Example": _native.FMT_EXAMPLE, "SequenceExample": _native.FMT_SEQUENCE}


def encode_batch(batch: RecordBatch, record_type: str) -> bytes:
    """RecordBatch -> framed TFRecord file image (uncompressed)."""
    if record_type == "ByteArray":
        col = batch.columns[0]
        return bytes(_native.frame_byte_arrays(
            np.ascontiguousarray(col.values, np.uint8),
            np.ascontiguousarray(col.elem_off, np.int64)))
    blob = schema_blob(batch.schema)
    dicts = [wire_to_native_dict(c) for c in batch.columns]
    return bytes(_native.encode_records(blob, FMT[record_type], dicts,
                                        batch.num_rows))


def decode_buffer(data: np.ndarray, schema: StructType, record_type: str,
                  verify_crc: bool = True) -> RecordBatch:
    """Framed TFRecord bytes -> RecordBatch (numpy wire-form)."""
    data = np.ascontiguousarray(data, np.uint8)
    off, lens = _native.scan_frames(data, verify_crc)
    R = len(off)
    if record_type == "ByteArray":
        # payload extents ARE the binary column (elem per row); vectorized
        # ragged gather instead of a python loop over records
        from ..arrow_interop import _ragged_gather_idx

        starts = np.asarray(off, np.int64)
        sizes = np.asarray(lens, np.int64)
        elem_off = np.zeros(R + 1, np.int64)
        np.cumsum(sizes, out=elem_off[1:])
        out = data[_ragged_gather_idx(starts, sizes)]
        col = WireColumn(kind=1, is_seq=False,
                         presence=np.ones(R, np.uint8),
                         row_off=np.arange(R + 1, dtype=np.int64),
                         values=out, elem_off=elem_off)
        return RecordBatch(schema, [col], R)
    blob = schema_blob(schema)
    dicts = _native.decode_records(data, off, lens, blob, FMT[record_type])
    cols = []
    for f, d in zip(wire_fields(schema), dicts):
        from ..schema import is_sequence_field, wire_kind_of
        cols.append(native_dict_to_wire(d, wire_kind_of(f.dataType),
                                        is_sequence_field(f.dataType)))
    return RecordBatch(schema, cols, R)
