"""Schema inference.

Reference semantics (TensorFlowInferSchema.scala, DefaultSource.scala:31-70):
  - the schema comes from the FIRST non-empty file only (collectFirst), but
    that file is scanned fully;
  - per row: 0 elements -> null, 1 -> scalar, >1 -> array; FeatureLists always
    infer as Array(Array(T)); bytes infer as String;
  - types merge commutatively by the precedence lattice (schema.py);
  - all-null features become NullType fields;
  - ByteArray record type has the fixed schema
    StructType([StructField("byteArray", BinaryType)]).

The heavy per-record scan runs in native code (host) or as a GPU kernel; this
module merges the resulting {name: lattice_code} maps — a max-reduce, which
distributed mode runs as an RCCL all-reduce over aligned code vectors
(parallel/dist.py).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from . import _native
from .schema import (
    BinaryType,
    StructField,
    StructType,
    type_from_lattice_code,
)

__all__ = ["infer_codes_from_buffer", "merge_code_maps", "schema_from_codes",
           "byte_array_schema"]


def byte_array_schema() -> StructType:
    # Reference: TensorFlowInferSchema.scala:60-64
    return StructType([StructField("byteArray", BinaryType(), True)])


def infer_codes_from_buffer(data: np.ndarray, rec_off: np.ndarray,
                            rec_len: np.ndarray, record_type: str) -> Dict[str, int]:
    fmt = _native.FMT_SEQUENCE if record_type == "SequenceExample" else _native.FMT_EXAMPLE
    return dict(_native.infer_schema_codes(data, rec_off, rec_len, fmt))


def merge_code_maps(maps: List[Dict[str, int]]) -> Dict[str, int]:
    out: Dict[str, int] = {}
    for m in maps:
        for k, v in m.items():
            out[k] = max(out.get(k, 0), v)
    return out


def schema_from_codes(codes: Dict[str, int]) -> StructType:
    """Deterministic field order: sorted by name (the reference's Scala Map
    ordering is arbitrary; sorting makes runs reproducible)."""
    fields = [StructField(name, type_from_lattice_code(code), True)
              for name, code in sorted(codes.items())]
    return StructType(fields)
