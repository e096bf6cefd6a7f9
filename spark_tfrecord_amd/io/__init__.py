from . import paths
from .reader import read_tfrecord
from .writer import write_tfrecord

__all__ = ["paths", "read_tfrecord", "write_tfrecord"]
