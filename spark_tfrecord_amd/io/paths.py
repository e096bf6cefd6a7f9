"""File layout, codecs, save modes, and partition discovery.

Mirrors the on-disk conventions the reference inherits from Spark's
FileFormatWriter / commit protocol (SURVEY.md §3.3, §5 "Failure detection"):
  - output dir contains part files `part-<shard>-<uuid>.tfrecord[.gz]`
  - `partitionBy` produces `col=value/` subdirectories
  - `_SUCCESS` marker written on job completion
  - save modes: error (default), overwrite, append, ignore
    (TFRecordIOSuite.scala:184-237 semantics)
  - `codec` option on write; read-side codec inferred from file extension
    (DefaultSource.scala:95-102)
Writes go to a temp file + atomic rename (idempotent shard retry).
"""

from __future__ import annotations

import glob as _glob
import gzip
import os
import shutil
import uuid
import zlib
from typing import Dict, List, Optional, Tuple

__all__ = ["normalize_codec", "codec_extension", "compress_bytes",
           "decompress_file", "list_data_files", "partition_values_of",
           "apply_save_mode", "part_file_name", "write_file_atomic",
           "write_success_marker", "SaveModeError", "hidden_tmp_path",
           "escape_path_name", "unescape_path_name"]

# codec option values accepted, mirroring Hadoop codec class names + shortcuts
_CODEC_ALIASES = {
    "gzip": "gzip",
    "org.apache.hadoop.io.compress.gzipcodec": "gzip",
    "deflate": "deflate",
    "org.apache.hadoop.io.compress.deflatecodec": "deflate",
    "org.apache.hadoop.io.compress.defaultcodec": "deflate",
    "none": None,
    "uncompressed": None,
}

_EXTENSIONS = {"gzip": ".gz", "deflate": ".deflate"}
_EXT_TO_CODEC = {".gz": "gzip", ".deflate": "deflate"}


class SaveModeError(RuntimeError):
    pass


def normalize_codec(codec: Optional[str]) -> Optional[str]:
    if codec is None:
        return None
    key = codec.strip().lower()
    if key in _CODEC_ALIASES:
        return _CODEC_ALIASES[key]
    raise ValueError(f"Unknown compression codec: {codec!r}")


def codec_extension(codec: Optional[str]) -> str:
    return _EXTENSIONS.get(codec, "")


def codec_from_path(path: str) -> Optional[str]:
    _, ext = os.path.splitext(path)
    return _EXT_TO_CODEC.get(ext)


# Gzip output is written as a SINGLE standard gzip member whose deflate
# stream carries Z_FULL_FLUSH sync points every _GZ_SEGMENT of input (each
# flush resets the dictionary and byte-aligns with the 00 00 FF FF empty
# stored block, pigz-style). Any gzip reader — TensorFlow included — decodes
# the file normally; OUR reader finds the sync markers and inflates the
# segments in parallel, verifying the member's CRC32 trailer (false-positive
# markers fall back to sequential inflate). Matches the reference's
# isSplitable=false model: gzip files still read as whole files, just on
# more than one core.
_GZ_SEGMENT = 4 << 20
_GZ_MARK = b"\x00\x00\xff\xff"


def compress_bytes(data: bytes, codec: Optional[str]) -> bytes:
    if codec is None:
        return data
    if codec == "gzip":
        c = zlib.compressobj(6, zlib.DEFLATED, 16 + 15)  # gzip wrapper
        out = []
        for pos in range(0, len(data), _GZ_SEGMENT):
            out.append(c.compress(data[pos:pos + _GZ_SEGMENT]))
            if pos + _GZ_SEGMENT < len(data):
                out.append(c.flush(zlib.Z_FULL_FLUSH))
        out.append(c.flush())
        return b"".join(out)
    if codec == "deflate":
        return zlib.compress(data, 6)
    raise ValueError(codec)


def _gunzip_parallel(raw: bytes) -> Optional[bytes]:
    """Parallel segmented inflate of a single-member gzip blob written with
    full-flush sync points. Returns None when the fast path does not apply
    (multi-member, extra header fields, marker false positive, CRC mismatch)."""
    if len(raw) < 20 or raw[:3] != b"\x1f\x8b\x08" or raw[3] != 0:
        return None  # flags would shift the header; let zlib handle it
    body = raw[10:-8]
    crc_want = int.from_bytes(raw[-8:-4], "little")
    isize = int.from_bytes(raw[-4:], "little")
    cuts = []
    p = body.find(_GZ_MARK)
    while p != -1:
        cuts.append(p + 4)
        p = body.find(_GZ_MARK, p + 4)
    if not cuts:
        return None
    bounds = [0] + cuts + [len(body)]
    segs = [body[bounds[i]:bounds[i + 1]] for i in range(len(bounds) - 1)]

    def inflate(seg):
        d = zlib.decompressobj(-15)
        return d.decompress(seg) + d.flush()

    from concurrent.futures import ThreadPoolExecutor
    try:
        with ThreadPoolExecutor(max_workers=min(16, os.cpu_count() or 4)) as ex:
            parts = list(ex.map(inflate, segs))
    except zlib.error:
        return None
    out = b"".join(parts)
    if len(out) % (1 << 32) != isize or (zlib.crc32(out) & 0xFFFFFFFF) != crc_want:
        return None
    return out


def decompress_file(path: str) -> bytes:
    codec = codec_from_path(path)
    with open(path, "rb") as f:
        raw = f.read()
    if codec is None:
        return raw
    if codec == "gzip":
        out = _gunzip_parallel(raw)
        return out if out is not None else gzip.decompress(raw)
    if codec == "deflate":
        return zlib.decompress(raw)
    raise ValueError(codec)


def part_file_name(shard: int, codec: Optional[str], job_id: str) -> str:
    return f"part-{shard:05d}-{job_id}.tfrecord{codec_extension(codec)}"


def hidden_tmp_path(final_path: str, suffix: str = "inprogress") -> str:
    """In-progress temp name for `final_path`, HIDDEN from readers: the
    basename gets a leading '.' so _is_data_file never lists a crashed
    job's leftovers as data (the analog of Spark's _temporary staging)."""
    d, base = os.path.split(final_path)
    return os.path.join(d, f".{base}.{suffix}")


# Temp-file suffixes from older layouts; excluded from reads as well so a
# partial file from a crashed pre-fix job is never scanned as data.
_TMP_MARKERS = (".inprogress", ".__tmp.")


def _is_data_file(name: str) -> bool:
    base = os.path.basename(name)
    if base.startswith("_") or base.startswith("."):
        return False
    return not any(m in base for m in _TMP_MARKERS)


def list_data_files(path) -> List[str]:
    """Resolve a path/glob/dir — or a list of them — into a sorted list of
    data files, recursing into partition directories."""
    if isinstance(path, (list, tuple)):
        out: List[str] = []
        for p in path:
            out.extend(list_data_files(p))
        return sorted(out)
    paths: List[str] = []
    candidates = _glob.glob(path) if _glob.has_magic(path) else [path]
    if _glob.has_magic(path) and not candidates:
        raise FileNotFoundError(f"Path does not exist: {path}")
    for p in candidates:
        if os.path.isdir(p):
            for root, dirs, files in os.walk(p):
                dirs[:] = sorted(d for d in dirs if _is_data_file(d))
                for fn in sorted(files):
                    if _is_data_file(fn):
                        paths.append(os.path.join(root, fn))
        elif os.path.isfile(p):
            paths.append(p)
        else:
            raise FileNotFoundError(f"Path does not exist: {p}")
    return sorted(paths)


# ---------------------------------------------------------------------------
# Hive-style partition path escaping. Spark escapes these characters in
# `col=value/` components via ExternalCatalogUtils.escapePathName (the layer
# above the reference library; its layout tests TFRecordIOSuite.scala:140-151
# rely on it). A value containing '/', '=' or '%' must not change the
# directory structure or collide with the escape syntax itself.
# ---------------------------------------------------------------------------

_NEEDS_ESCAPE = set('"#%\'*/:=?\\{[]^\x7f') | {chr(c) for c in range(0x20)}


def escape_path_name(value: str) -> str:
    """Escape a partition value for use as the `value` of a `col=value/`
    path component (Hive/Spark %XX escaping)."""
    out = []
    for ch in value:
        if ch in _NEEDS_ESCAPE:
            out.append(f"%{ord(ch):02X}")
        else:
            out.append(ch)
    return "".join(out)


def unescape_path_name(comp: str) -> str:
    """Inverse of escape_path_name (tolerates malformed % sequences)."""
    out = []
    i = 0
    n = len(comp)
    while i < n:
        ch = comp[i]
        if ch == "%" and i + 2 < n + 1 and i + 3 <= n:
            try:
                out.append(chr(int(comp[i + 1:i + 3], 16)))
                i += 3
                continue
            except ValueError:
                pass
        out.append(ch)
        i += 1
    return "".join(out)


def partition_values_of(file_path: str, base_dir: str) -> Dict[str, str]:
    """Extract `col=value` partition components between base_dir and the
    file, unescaping Hive-style %XX sequences in both name and value."""
    rel = os.path.relpath(os.path.dirname(os.path.abspath(file_path)),
                          os.path.abspath(base_dir))
    out: Dict[str, str] = {}
    if rel in (".", ""):
        return out
    for comp in rel.split(os.sep):
        if "=" in comp:
            k, v = comp.split("=", 1)
            out[unescape_path_name(k)] = unescape_path_name(v)
    return out


def apply_save_mode(path: str, mode: str) -> bool:
    """Prepare the output dir for the given save mode.

    Returns True when the write should proceed, False for ignore-and-skip.
    Mirrors Spark SaveMode semantics exercised in TFRecordIOSuite.scala:184-237.
    """
    mode = mode.lower().replace("_", "")
    exists = os.path.exists(path) and (not os.path.isdir(path) or os.listdir(path))
    if mode in ("error", "errorifexists", "default"):
        if exists:
            raise SaveModeError(f"path already exists: {path}")
    elif mode == "overwrite":
        if os.path.isdir(path):
            shutil.rmtree(path)
        elif os.path.exists(path):
            os.remove(path)
    elif mode == "append":
        pass
    elif mode == "ignore":
        if exists:
            return False
    else:
        raise ValueError(f"Unknown save mode: {mode}")
    os.makedirs(path, exist_ok=True)
    return True


def write_file_atomic(data: bytes, final_path: str):
    """Temp file + rename: a torn write never becomes visible (the engine's
    stand-in for Spark's task-commit rename protocol). The temp name is
    dot-prefixed so a crashed job's leftover is never listed as data."""
    os.makedirs(os.path.dirname(final_path), exist_ok=True)
    tmp = hidden_tmp_path(final_path, f"tmp.{uuid.uuid4().hex[:8]}")
    with open(tmp, "wb") as f:
        f.write(data)
        f.flush()
        os.fsync(f.fileno())
    os.replace(tmp, final_path)


def write_success_marker(path: str):
    with open(os.path.join(path, "_SUCCESS"), "wb"):
        pass
