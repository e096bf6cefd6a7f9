"""Schema type system for the MI355X-native TFRecord engine.

Mirrors the Catalyst data types that the reference library supports
(reference: TFRecordSerializer.scala:68-152, TFRecordDeserializer.scala:68-124)
and the schema-inference type lattice
(reference: TensorFlowInferSchema.scala:194-228).

The types here are deliberately a small, self-contained value-object layer:
the engine's on-device representation is columnar (see `columnar.py`), and
these objects only describe logical shape + nullability the way a Spark
StructType would.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Union

__all__ = [
    "DataType",
    "NullType",
    "IntegerType",
    "LongType",
    "FloatType",
    "DoubleType",
    "DecimalType",
    "StringType",
    "BinaryType",
    "ArrayType",
    "StructField",
    "StructType",
    "KIND_BYTES",
    "KIND_FLOAT",
    "KIND_INT64",
    "wire_kind_of",
    "is_sequence_field",
    "validate_schema_for_record_type",
    "merge_types",
    "type_precedence",
]

# Wire "kind" codes == the protobuf Feature oneof field numbers
# (reference: TensorFlowInferSchema.scala:133-141 uses
#  Feature.BYTES_LIST_FIELD_NUMBER=1, FLOAT_LIST=2, INT64_LIST=3).
KIND_BYTES = 1
KIND_FLOAT = 2
KIND_INT64 = 3


class DataType:
    """Base class for all logical data types."""

    def simple_string(self) -> str:
        return type(self).__name__.replace("Type", "").lower()

    def __repr__(self) -> str:
        return f"{type(self).__name__}()"

    def __eq__(self, other) -> bool:
        return type(self) is type(other)

    def __hash__(self) -> int:
        return hash(type(self).__name__)


class NullType(DataType):
    pass


class IntegerType(DataType):
    pass


class LongType(DataType):
    pass


class FloatType(DataType):
    pass


class DoubleType(DataType):
    pass


@dataclass(frozen=True)
class DecimalType(DataType):
    """Decimal — serialized as float32 like the reference
    (TFRecordSerializer.scala:88-90)."""

    precision: int = 38
    scale: int = 18

    def simple_string(self) -> str:
        return f"decimal({self.precision},{self.scale})"

    def __eq__(self, other):
        return (
            isinstance(other, DecimalType)
            and other.precision == self.precision
            and other.scale == self.scale
        )

    def __hash__(self):
        return hash(("decimal", self.precision, self.scale))


class StringType(DataType):
    pass


class BinaryType(DataType):
    pass


@dataclass(frozen=True)
class ArrayType(DataType):
    elementType: DataType = field(default_factory=LongType)
    containsNull: bool = True

    def simple_string(self) -> str:
        return f"array<{self.elementType.simple_string()}>"

    def __eq__(self, other):
        return isinstance(other, ArrayType) and other.elementType == self.elementType

    def __hash__(self):
        return hash(("array", self.elementType))

    def __repr__(self):
        return f"ArrayType({self.elementType!r})"


@dataclass
class StructField:
    name: str
    dataType: DataType
    nullable: bool = True

    def __repr__(self):
        return f"StructField({self.name!r}, {self.dataType!r}, nullable={self.nullable})"


class StructType(DataType):
    def __init__(self, fields: Optional[List[StructField]] = None):
        self.fields: List[StructField] = list(fields or [])

    def add(self, name_or_field: Union[str, StructField], dataType: Optional[DataType] = None,
            nullable: bool = True) -> "StructType":
        if isinstance(name_or_field, StructField):
            self.fields.append(name_or_field)
        else:
            self.fields.append(StructField(name_or_field, dataType, nullable))
        return self

    @property
    def names(self) -> List[str]:
        return [f.name for f in self.fields]

    def __getitem__(self, key):
        if isinstance(key, int):
            return self.fields[key]
        for f in self.fields:
            if f.name == key:
                return f
        raise KeyError(key)

    def __contains__(self, name) -> bool:
        return any(f.name == name for f in self.fields)

    def __iter__(self):
        return iter(self.fields)

    def __len__(self):
        return len(self.fields)

    def __eq__(self, other):
        return isinstance(other, StructType) and [
            (f.name, f.dataType) for f in self.fields
        ] == [(f.name, f.dataType) for f in other.fields]

    def __hash__(self):
        return hash(tuple((f.name, f.dataType) for f in self.fields))

    def __repr__(self):
        inner = ", ".join(repr(f) for f in self.fields)
        return f"StructType([{inner}])"

    def simple_string(self) -> str:
        inner = ",".join(f"{f.name}:{f.dataType.simple_string()}" for f in self.fields)
        return f"struct<{inner}>"


# ---------------------------------------------------------------------------
# Wire-kind mapping (serializer side).
# Reference: TFRecordSerializer.scala:68-152 —
#   Int/Long -> Int64List; Float/Double/Decimal -> FloatList (float32);
#   String/Binary -> BytesList; arrays map by element type;
#   ArrayType(ArrayType(_)) -> SequenceExample FeatureList.
# ---------------------------------------------------------------------------

_SCALAR_KIND = {
    IntegerType: KIND_INT64,
    LongType: KIND_INT64,
    FloatType: KIND_FLOAT,
    DoubleType: KIND_FLOAT,
    DecimalType: KIND_FLOAT,
    StringType: KIND_BYTES,
    BinaryType: KIND_BYTES,
}


def _scalar_kind(dt: DataType) -> int:
    for cls, kind in _SCALAR_KIND.items():
        if isinstance(dt, cls):
            return kind
    raise TypeError(f"Cannot convert data type {dt!r} to a TFRecord feature kind")


def wire_kind_of(dt: DataType) -> int:
    """Feature kind (1=bytes, 2=float, 3=int64) for a field's logical type."""
    if isinstance(dt, ArrayType):
        inner = dt.elementType
        if isinstance(inner, ArrayType):
            return _scalar_kind(inner.elementType)
        return _scalar_kind(inner)
    if isinstance(dt, NullType):
        # Nothing is ever emitted for a NullType column; int64 is a harmless
        # placeholder for descriptor purposes.
        return KIND_INT64
    return _scalar_kind(dt)


def is_sequence_field(dt: DataType) -> bool:
    """True when the field maps to a SequenceExample FeatureList
    (2-D ragged), i.e. ArrayType(ArrayType(_)).
    Reference: TFRecordSerializer.scala:45-47."""
    return isinstance(dt, ArrayType) and isinstance(dt.elementType, ArrayType)


def validate_schema_for_record_type(schema: "StructType", record_type: str):
    """Reject schemas the chosen record type cannot carry, like the
    reference's converter construction does (TFRecordSerializer.scala:147-180
    raises for types unsupported by the record type): a 2-D ragged column
    (SequenceExample FeatureList) has no representation in an Example —
    silently dropping it would be data loss.
    """
    if record_type != "Example":
        return
    for f in schema.fields:
        if is_sequence_field(f.dataType):
            raise TypeError(
                f"Cannot convert field '{f.name}' of type "
                f"{f.dataType.simple_string()} with recordType 'Example': "
                f"nested arrays require recordType 'SequenceExample'")


# ---------------------------------------------------------------------------
# Schema-inference type lattice.
# Reference: TensorFlowInferSchema.scala:194-228 (getNumericPrecedence /
# findTightestCommonType): Long < Float < String < Arr[Long] < Arr[Float]
# < Arr[Str] < Arr[Arr[Long]] < Arr[Arr[Float]] < Arr[Arr[Str]].
# NullType merges to the other side's type.
# ---------------------------------------------------------------------------

_LATTICE: List[DataType] = [
    LongType(),
    FloatType(),
    StringType(),
    ArrayType(LongType()),
    ArrayType(FloatType()),
    ArrayType(StringType()),
    ArrayType(ArrayType(LongType())),
    ArrayType(ArrayType(FloatType())),
    ArrayType(ArrayType(StringType())),
]


def type_precedence(dt: DataType) -> int:
    """Index in the promotion lattice; raises for types outside it."""
    for i, t in enumerate(_LATTICE):
        if t == dt:
            return i
    raise TypeError(f"Unsupported type in schema inference lattice: {dt!r}")


def merge_types(a: Optional[DataType], b: Optional[DataType]) -> Optional[DataType]:
    """Commutative merge of two inferred types (lattice max).

    Mirrors findTightestCommonType (TensorFlowInferSchema.scala:213-228):
    equal -> same; null/None -> other; otherwise the higher-precedence type.
    """
    if a is None or isinstance(a, NullType):
        return b
    if b is None or isinstance(b, NullType):
        return a
    if a == b:
        return a
    pa, pb = type_precedence(a), type_precedence(b)
    return _LATTICE[max(pa, pb)]


def lattice_code(dt: Optional[DataType]) -> int:
    """Encode an inferred type as a small int for RCCL max-all-reduce.

    0 = null/absent; 1..9 = lattice precedence + 1. The merge of two codes is
    their max, which makes distributed schema inference a plain all-reduce
    (lattice max is commutative/associative => reduce-safe).
    """
    if dt is None or isinstance(dt, NullType):
        return 0
    return type_precedence(dt) + 1


def type_from_lattice_code(code: int) -> DataType:
    if code == 0:
        return NullType()
    return _LATTICE[code - 1]
