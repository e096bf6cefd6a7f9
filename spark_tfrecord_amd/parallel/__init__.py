from .dist import (
    infer_schema_distributed,
    init_distributed,
    read_tfrecord_distributed,
    shard_files,
    write_tfrecord_distributed,
)

__all__ = ["init_distributed", "shard_files", "infer_schema_distributed",
           "write_tfrecord_distributed", "read_tfrecord_distributed"]
