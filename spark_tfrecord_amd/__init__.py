"""spark_tfrecord_amd — an MI355X-native TFRecord I/O engine.

A from-scratch re-design of linkedin/spark-tfrecord's capabilities
(format("tfrecord"), recordType = Example | SequenceExample | ByteArray,
schema inference with the reference's type-promotion lattice, partitionBy
writes, save modes, gzip codecs, identical on-disk bytes) with the hot path
in C++/HIP for CDNA4 (gfx950) and multi-GPU scaling over RCCL/xGMI instead
of a JVM + Spark. See SURVEY.md for the structural map of the reference.
"""

# torch must load first so _native.so binds the same (torch-bundled) HIP
# runtime: loading the system libamdhip64 first leaves torch and _native on
# two different runtimes, and kernel launches fail with hipErrorNoDevice.
import torch  # noqa: F401  (intentional import order)

from .api import DataFrame, DataFrameReader, DataFrameWriter, TFRecordSession, session
from .io.reader import count_tfrecord, read_tfrecord
from .io.writer import write_tfrecord
from .io.stream_writer import ShardWriter
from .io.validate import validate_tfrecord
from . import torch_data
from .schema import (
    ArrayType,
    BinaryType,
    DataType,
    DecimalType,
    DoubleType,
    FloatType,
    IntegerType,
    LongType,
    NullType,
    StringType,
    StructField,
    StructType,
)

__version__ = "0.1.0"

__all__ = [
    "session", "DataFrame", "DataFrameReader", "DataFrameWriter",
    "TFRecordSession", "read_tfrecord", "write_tfrecord", "validate_tfrecord",
    "count_tfrecord", "ShardWriter",
    "DataType", "NullType", "IntegerType", "LongType", "FloatType",
    "DoubleType", "DecimalType", "StringType", "BinaryType", "ArrayType",
    "StructField", "StructType",
]
