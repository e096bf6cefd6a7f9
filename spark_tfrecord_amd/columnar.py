"""Columnar wire-form batches and serde between user data and the native codec.

The engine's canonical in-memory representation is an Arrow-like ragged
columnar layout ("wire-form") shared byte-for-byte with csrc/codec_core.h:

    non-seq field : presence u8[R]; row_off i64[R+1] (values per row, cumulative)
    seq field     : + list_off i64[R+1] (sub-lists per row), sub_off i64[L+1]
    bytes kind    : values = flat u8; elem_off i64[E+1] (bytes per string)
    int64 / float : values = i64[V] / f32[V]

This mirrors the reference's serializer/deserializer semantics
(TFRecordSerializer.scala:68-180, TFRecordDeserializer.scala:68-175):
scalars are single-element lists on the wire, Double/Decimal are downcast to
float32 on write, scalar reads take the head element, Integer reads downcast
int64, and null handling follows nullability (omit vs error).
"""

from __future__ import annotations

import decimal
import struct
from dataclasses import dataclass, field
from typing import Any, List, Optional, Sequence

import numpy as np

from .schema import (
    ArrayType,
    BinaryType,
    DataType,
    DecimalType,
    DoubleType,
    FloatType,
    IntegerType,
    KIND_BYTES,
    KIND_FLOAT,
    KIND_INT64,
    LongType,
    NullType,
    StringType,
    StructField,
    StructType,
    is_sequence_field,
    wire_kind_of,
)

__all__ = ["WireColumn", "RecordBatch", "schema_blob", "column_from_values",
           "column_to_pylist", "wire_to_native_dict", "native_dict_to_wire"]


def schema_blob(schema: StructType) -> bytes:
    """Serialize a schema into the flat blob csrc/codec_core.h::schema_view
    reads (identical bytes go to host calls and device constant buffers)."""
    names = b""
    descs = b""
    nf = 0
    for f in schema.fields:
        if isinstance(f.dataType, NullType):
            continue  # NullType columns never touch the wire
        nb = f.name.encode("utf-8")
        kind = wire_kind_of(f.dataType)
        seq = 1 if is_sequence_field(f.dataType) else 0
        descs += struct.pack("<iiii", kind, seq, len(names), len(nb))
        names += nb
        nf += 1
    return struct.pack("<i", nf) + descs + names


def wire_fields(schema: StructType) -> List[StructField]:
    """Fields that participate in the wire blob (NullType columns excluded),
    in blob order."""
    return [f for f in schema.fields if not isinstance(f.dataType, NullType)]


@dataclass
class WireColumn:
    """One field's ragged wire-form buffers (numpy on CPU, torch on GPU)."""

    kind: int
    is_seq: bool
    presence: Any            # u8[R]
    row_off: Any             # i64[R+1]
    values: Any              # i64[V] | f32[V] | u8[B]
    elem_off: Optional[Any] = None   # i64[E+1], bytes kind
    list_off: Optional[Any] = None   # i64[R+1], seq
    sub_off: Optional[Any] = None    # i64[L+1], seq

    @property
    def num_rows(self) -> int:
        return len(self.presence)


@dataclass
class RecordBatch:
    """A batch of rows: schema + one WireColumn per wire field."""

    schema: StructType
    columns: List[WireColumn]
    num_rows: int

    def column(self, name: str) -> WireColumn:
        for f, c in zip(wire_fields(self.schema), self.columns):
            if f.name == name:
                return c
        raise KeyError(name)


# ---------------------------------------------------------------------------
# native-dict <-> WireColumn (the pybind boundary)
# ---------------------------------------------------------------------------

def wire_to_native_dict(col: WireColumn) -> dict:
    d = {
        "presence": np.ascontiguousarray(col.presence, dtype=np.uint8),
        "row_off": np.ascontiguousarray(col.row_off, dtype=np.int64),
    }
    if col.kind == KIND_INT64:
        d["values_i64"] = np.ascontiguousarray(col.values, dtype=np.int64)
    elif col.kind == KIND_FLOAT:
        d["values_f32"] = np.ascontiguousarray(col.values, dtype=np.float32)
    else:
        d["values_bytes"] = np.ascontiguousarray(col.values, dtype=np.uint8)
        d["elem_off"] = np.ascontiguousarray(col.elem_off, dtype=np.int64)
    if col.is_seq:
        d["list_off"] = np.ascontiguousarray(col.list_off, dtype=np.int64)
        d["sub_off"] = np.ascontiguousarray(col.sub_off, dtype=np.int64)
    return d


def native_dict_to_wire(d: dict, kind: int, is_seq: bool) -> WireColumn:
    """Wrap the decoder's per-field output dict. elem_len/sub_count are raw
    per-element counts from the kernels; prefix-sum them here."""
    elem_off = None
    if "elem_len" in d:
        elem_off = np.zeros(len(d["elem_len"]) + 1, dtype=np.int64)
        np.cumsum(d["elem_len"], out=elem_off[1:])
    sub_off = None
    if "sub_count" in d:
        sub_off = np.zeros(len(d["sub_count"]) + 1, dtype=np.int64)
        np.cumsum(d["sub_count"], out=sub_off[1:])
    return WireColumn(
        kind=kind,
        is_seq=is_seq,
        presence=d["presence"],
        row_off=d["row_off"],
        values=d["values"],
        elem_off=elem_off,
        list_off=d.get("list_off"),
        sub_off=sub_off,
    )


# ---------------------------------------------------------------------------
# Serializer side: python values -> WireColumn
# (converter semantics of TFRecordSerializer.scala:68-180)
# ---------------------------------------------------------------------------

def _to_f32(v, name: str):
    if isinstance(v, decimal.Decimal):
        return np.float32(float(v))
    return np.float32(v)  # Double -> float32 downcast, by design (lossy)


def _to_i64(v, name: str):
    if isinstance(v, (bool, np.bool_)):
        raise TypeError(f"Cannot convert field '{name}': boolean is not supported")
    return np.int64(v)


def _to_bytes(v, name: str) -> bytes:
    if isinstance(v, str):
        return v.encode("utf-8")
    if isinstance(v, (bytes, bytearray, memoryview)):
        return bytes(v)
    if isinstance(v, np.ndarray) and v.dtype == np.uint8:
        return v.tobytes()
    raise TypeError(f"Cannot convert field '{name}' value {type(v).__name__} to bytes")


def _validate_serializable(dt: DataType, name: str):
    """Reject unsupported types at construction time, like the serializer's
    constructor does (TFRecordSerializer.scala:147-151)."""
    if isinstance(dt, NullType):
        return
    if isinstance(dt, ArrayType):
        inner = dt.elementType
        if isinstance(inner, ArrayType):
            if isinstance(inner.elementType, (ArrayType, StructType, NullType)):
                raise TypeError(
                    f"Cannot convert field '{name}': arrays nested deeper than 2 are not supported")
            wire_kind_of(inner.elementType)
            return
        if isinstance(inner, (StructType, NullType)):
            raise TypeError(f"Cannot convert field '{name}' to a TFRecord feature")
        wire_kind_of(inner)
        return
    if isinstance(dt, StructType):
        raise TypeError(f"Cannot convert field '{name}': nested structs are not supported")
    wire_kind_of(dt)  # raises TypeError for anything else unsupported


def _is_null(v) -> bool:
    if v is None:
        return True
    # pandas NaN / NaT for object columns
    if isinstance(v, float) and v != v:
        return True
    return False


def column_from_values(values: Sequence, dt: DataType, nullable: bool,
                       name: str) -> WireColumn:
    """Build a WireColumn from a sequence of python values (one per row).

    Null handling mirrors serializeExample (TFRecordSerializer.scala:20-35):
    null + nullable => feature omitted; null + non-nullable => error.
    """
    _validate_serializable(dt, name)
    kind = wire_kind_of(dt)
    seq = is_sequence_field(dt)
    R = len(values)

    # numpy fast path for numeric scalar columns with no nulls
    if not seq and not isinstance(dt, ArrayType) and isinstance(values, np.ndarray) \
            and values.dtype != object:
        if kind == KIND_INT64 and np.issubdtype(values.dtype, np.integer):
            vals = values.astype(np.int64, copy=False)
            return WireColumn(kind, False, np.ones(R, np.uint8),
                              np.arange(R + 1, dtype=np.int64), vals)
        if kind == KIND_FLOAT and np.issubdtype(values.dtype, np.floating):
            vals = values.astype(np.float32, copy=False)
            return WireColumn(kind, False, np.ones(R, np.uint8),
                              np.arange(R + 1, dtype=np.int64), vals)

    presence = np.ones(R, np.uint8)
    row_counts = np.zeros(R, np.int64)
    flat: List[Any] = []
    list_counts = np.zeros(R, np.int64) if seq else None
    sub_counts: List[int] = []

    for r, v in enumerate(values):
        if _is_null(v):
            if not nullable:
                raise ValueError(
                    f"null value in non-nullable field '{name}' (row {r})")
            presence[r] = 0
            continue
        if seq:
            n = 0
            nls = 0
            for sub in v:
                if sub is None:
                    raise ValueError(
                        f"null inner array in field '{name}' (row {r})")
                sub = list(sub)
                sub_counts.append(len(sub))
                flat.extend(sub)
                n += len(sub)
                nls += 1
            row_counts[r] = n
            list_counts[r] = nls
        elif isinstance(dt, ArrayType):
            v = list(v)
            flat.extend(v)
            row_counts[r] = len(v)
        else:
            flat.append(v)
            row_counts[r] = 1

    row_off = np.zeros(R + 1, np.int64)
    np.cumsum(row_counts, out=row_off[1:])

    elem_off = None
    if kind == KIND_INT64:
        vals = np.fromiter((_to_i64(v, name) for v in flat), np.int64, len(flat))
    elif kind == KIND_FLOAT:
        vals = np.fromiter((_to_f32(v, name) for v in flat), np.float32, len(flat))
    else:
        bs = [_to_bytes(v, name) for v in flat]
        elem_off = np.zeros(len(bs) + 1, np.int64)
        np.cumsum(np.fromiter((len(b) for b in bs), np.int64, len(bs)),
                  out=elem_off[1:])
        vals = np.frombuffer(b"".join(bs), np.uint8).copy() if bs else np.zeros(0, np.uint8)

    list_off = sub_off = None
    if seq:
        list_off = np.zeros(R + 1, np.int64)
        np.cumsum(list_counts, out=list_off[1:])
        sub_off = np.zeros(len(sub_counts) + 1, np.int64)
        np.cumsum(np.asarray(sub_counts, np.int64), out=sub_off[1:])

    return WireColumn(kind, seq, presence, row_off, vals, elem_off, list_off, sub_off)


# ---------------------------------------------------------------------------
# Deserializer side: WireColumn -> python values
# (writer semantics of TFRecordDeserializer.scala:68-175)
# ---------------------------------------------------------------------------

def _scalar_convert(dt: DataType, raw):
    if isinstance(dt, IntegerType):
        return int(np.int32(raw))  # int64 -> int downcast, reference behavior
    if isinstance(dt, LongType):
        return int(raw)
    if isinstance(dt, FloatType):
        return float(np.float32(raw))
    if isinstance(dt, DoubleType):
        return float(raw)
    if isinstance(dt, DecimalType):
        return decimal.Decimal(repr(float(np.float32(raw))))
    raise TypeError(f"Unsupported scalar type {dt!r}")


def _bytes_convert(dt: DataType, b: bytes):
    if isinstance(dt, StringType):
        # replacement chars on invalid UTF-8, like protobuf-java's toStringUtf8
        return b.decode("utf-8", errors="replace")
    if isinstance(dt, BinaryType):
        return b
    raise TypeError(f"Unsupported bytes type {dt!r}")


def _check_kind(col: WireColumn, dt: DataType, name: str):
    expect = wire_kind_of(dt)
    if col.kind != expect:
        raise RuntimeError(
            f"Feature '{name}' kind does not match requested type {dt.simple_string()}")


def column_to_pylist(col: WireColumn, dt: DataType, nullable: bool,
                     name: str) -> List[Any]:
    """Materialize a WireColumn into python objects per the requested logical
    type. Missing + non-nullable raises (TFRecordDeserializer.scala:31,56);
    scalar reads take the head element; Integer downcasts."""
    if isinstance(dt, NullType):
        return [None] * col.num_rows
    _check_kind(col, dt, name)
    R = col.num_rows
    presence = np.asarray(col.presence)
    row_off = np.asarray(col.row_off)
    out: List[Any] = [None] * R

    def elem(vi: int):
        if col.kind == KIND_BYTES:
            b0, b1 = int(col.elem_off[vi]), int(col.elem_off[vi + 1])
            return bytes(np.asarray(col.values[b0:b1]).tobytes())
        return col.values[vi]

    seq = is_sequence_field(dt)
    if seq != col.is_seq:
        raise RuntimeError(f"Feature '{name}' dimensionality does not match schema")

    for r in range(R):
        if not presence[r]:
            if not nullable:
                raise ValueError(f"Feature '{name}' is required but missing (row {r})")
            out[r] = None
            continue
        if seq:
            inner_dt = dt.elementType.elementType
            lists = []
            for j in range(int(col.list_off[r]), int(col.list_off[r + 1])):
                v0, v1 = int(col.sub_off[j]), int(col.sub_off[j + 1])
                if col.kind == KIND_BYTES:
                    lists.append([_bytes_convert(inner_dt, elem(v)) for v in range(v0, v1)])
                else:
                    lists.append([_scalar_convert(inner_dt, col.values[v]) for v in range(v0, v1)])
            out[r] = lists
        elif isinstance(dt, ArrayType):
            v0, v1 = int(row_off[r]), int(row_off[r + 1])
            inner_dt = dt.elementType
            if col.kind == KIND_BYTES:
                out[r] = [_bytes_convert(inner_dt, elem(v)) for v in range(v0, v1)]
            else:
                out[r] = [_scalar_convert(inner_dt, col.values[v]) for v in range(v0, v1)]
        else:
            v0, v1 = int(row_off[r]), int(row_off[r + 1])
            if v1 == v0:
                # present feature with zero elements: scalar read has no head
                raise ValueError(
                    f"Feature '{name}' is present but empty; cannot read scalar (row {r})")
            if col.kind == KIND_BYTES:
                out[r] = _bytes_convert(dt, elem(v0))
            else:
                out[r] = _scalar_convert(dt, col.values[v0])
    return out
