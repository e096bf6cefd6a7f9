"""Spark-like fluent API.

The reference's entire public surface is the Spark DataFrameReader/Writer
contract plus two options (`recordType`, `codec`) and the
format name "tfrecord". This module mirrors that surface without a JVM:

    import spark_tfrecord_amd as stf
    df = stf.session.read.format("tfrecord") \
            .option("recordType", "SequenceExample").load(path)
    df.write.format("tfrecord").partitionBy("part") \
            .option("codec", "org.apache.hadoop.io.compress.GzipCodec") \
            .mode("overwrite").save(out)

DataFrame wraps a pyarrow Table (with .to_pandas() / .to_arrow_table() /
.collect() accessors and torch-tensor access for the GPU pipeline).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import pyarrow as pa

from .arrow_interop import schema_from_arrow
from .schema import StructType

__all__ = ["DataFrame", "DataFrameReader", "DataFrameWriter", "TFRecordSession",
           "session"]


class DataFrame:
    """A thin immutable DataFrame: pyarrow Table + engine schema."""

    def __init__(self, table: pa.Table, schema: Optional[StructType] = None):
        self._table = table
        self.schema = schema if schema is not None else schema_from_arrow(table.schema)

    # -- data access ------------------------------------------------------
    def to_arrow_table(self) -> pa.Table:
        return self._table

    def to_pandas(self):
        return self._table.to_pandas()

    def collect(self) -> List[dict]:
        return self._table.to_pylist()

    def count(self) -> int:
        return self._table.num_rows

    @property
    def num_rows(self) -> int:
        return self._table.num_rows

    @property
    def columns(self) -> List[str]:
        return list(self._table.column_names)

    def select(self, *cols: str) -> "DataFrame":
        names = list(cols)
        return DataFrame(self._table.select(names),
                         StructType([self.schema[c] for c in names]))

    def sort(self, *cols: str) -> "DataFrame":
        return DataFrame(self._table.sort_by([(c, "ascending") for c in cols]),
                         self.schema)

    def __getitem__(self, name: str):
        return self._table.column(name)

    def __len__(self) -> int:
        return self._table.num_rows

    def __repr__(self):
        return f"DataFrame[{self.schema.simple_string()}] ({self.num_rows} rows)"

    # -- write ------------------------------------------------------------
    @property
    def write(self) -> "DataFrameWriter":
        return DataFrameWriter(self)


class _OptionsMixin:
    _options: Dict[str, str]

    def option(self, key: str, value):
        self._options[key.lower()] = value
        return self

    def options(self, **kwargs):
        for k, v in kwargs.items():
            self.option(k, v)
        return self

    def format(self, fmt: str):
        if fmt.lower() not in ("tfrecord", "tfrecords"):
            raise ValueError(f"Unsupported format {fmt!r}: this engine serves "
                             "format('tfrecord')")
        return self


class DataFrameReader(_OptionsMixin):
    def __init__(self):
        self._options = {}
        self._schema: Optional[StructType] = None

    def schema(self, s: StructType):
        self._schema = s
        return self

    def load(self, path: str) -> DataFrame:
        from .io.reader import read_tfrecord

        return read_tfrecord(
            path,
            schema=self._schema,
            record_type=self._options.get("recordtype", "Example"),
            engine=self._options.get("engine", "auto"),
        )


class DataFrameWriter(_OptionsMixin):
    def __init__(self, df: DataFrame):
        self._df = df
        self._options = {}
        self._mode = "errorifexists"
        self._partition_by: Optional[Sequence[str]] = None

    def mode(self, m: str):
        self._mode = m
        return self

    def partitionBy(self, *cols: str):
        self._partition_by = [c for group in cols
                              for c in (group if isinstance(group, (list, tuple))
                                        else [group])]
        return self

    partition_by = partitionBy

    def save(self, path: str) -> None:
        from .io.writer import write_tfrecord

        write_tfrecord(
            self._df,
            path,
            record_type=self._options.get("recordtype", "Example"),
            codec=self._options.get("codec"),
            mode=self._mode,
            partition_by=self._partition_by,
            schema=self._df.schema,
            num_shards=int(self._options.get("num_shards", 1)),
            engine=self._options.get("engine", "auto"),
        )


class TFRecordSession:
    """Stand-in for the SparkSession entry point."""

    @property
    def read(self) -> DataFrameReader:
        return DataFrameReader()

    def createDataFrame(self, data, schema: Optional[StructType] = None) -> DataFrame:
        from .io.writer import normalize_input

        table = normalize_input(data, schema)
        return DataFrame(table, schema)


session = TFRecordSession()
