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


_POOL = None


def shared_pool():
    """Process-wide IO thread pool (compress/inflate/write workers). One
    persistent pool: per-call ThreadPoolExecutor spin-up dominated small
    round-trips (~1.2 ms of lock/thread-start per 1k-row write+read)."""
    global _POOL
    if _POOL is None:
        from concurrent.futures import ThreadPoolExecutor

        _POOL = ThreadPoolExecutor(max_workers=min(32, os.cpu_count() or 8),
                                   thread_name_prefix="tfrec-io")
    return _POOL


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
# stored block, pigz-style), and whose header carries an FEXTRA subfield
# ('T','S') listing every segment's (compressed, uncompressed) length.
# Any gzip reader — TensorFlow included — decodes the file normally (extra
# fields are skipped per RFC 1952); OUR readers get exact segment extents:
# the host inflates segments on a thread pool, the GPU engine inflates all
# segments in one kernel launch (one segment per lane, csrc/hip/inflate.hip)
# with output offsets known up front. Table-less gzip (foreign files, or
# > _GZ_MAX_SEGS segments) falls back to the marker scan, then to
# sequential inflate. Matches the reference's isSplitable=false model:
# gzip files still read as whole files, just on more than one core/CU.
# 32 KiB segments: device inflation is one segment per LANE and Huffman
# decode is bit-serial, so wall time ~= segment size — smaller segments buy
# parallelism AND a lower single-file latency floor directly. Compression-
# ratio cost of the extra dictionary resets is a few % at level 6
# (TFREC_GZ_SEGMENT overrides for ratio-sensitive datasets).
_GZ_SEGMENT = int(os.environ.get("TFREC_GZ_SEGMENT", 32 << 10))
# Segments whose level-6 deflate saves less than this ratio re-emit as
# STORED blocks (zstd-style incompressibility bailout, still standard
# gzip): decode streams at copy speed instead of one bit-serial Huffman
# symbol per byte. <= 0 disables.
_GZ_STORED_THRESHOLD = float(os.environ.get("TFREC_GZ_STORED_THRESHOLD",
                                            "0.98"))
_GZ_MARK = b"\x00\x00\xff\xff"
_GZ_MAX_SEGS = 8189  # FEXTRA payload cap: 65535 bytes / 8 per segment


def _gz_segment_size(total: int) -> int:
    """Segment size: the knob, grown so the table fits FEXTRA's 64 KiB."""
    seg = _GZ_SEGMENT
    while total > seg * _GZ_MAX_SEGS:
        seg *= 2
    return seg


def compress_bytes(data: bytes, codec: Optional[str]) -> bytes:
    if codec is None:
        return data
    if codec == "gzip":
        seg = _gz_segment_size(len(data))
        c = zlib.compressobj(6, zlib.DEFLATED, -15)  # raw deflate body
        chunks: List[bytes] = []
        seg_lens: List[Tuple[int, int]] = []  # (comp_len, uncomp_len)
        n = len(data)
        pos = 0
        while True:
            hi = min(pos + seg, n)
            body = c.compress(data[pos:hi])
            body += (c.flush() if hi == n else c.flush(zlib.Z_FULL_FLUSH))
            if len(body) >= _GZ_STORED_THRESHOLD * (hi - pos) > 0:
                # effectively incompressible: entropy-coding buys almost
                # nothing but costs ~30x on decode (every output byte
                # becomes a bit-serial Huffman symbol; stored segments
                # stream at copy speed on host AND device). Re-emit this
                # segment as stored blocks (level 0 — still standard
                # deflate; the full flush resets the dictionary, so
                # segments are independent and the swap is local).
                c0 = zlib.compressobj(0, zlib.DEFLATED, -15)
                body = c0.compress(data[pos:hi])
                body += (c0.flush() if hi == n else c0.flush(zlib.Z_FULL_FLUSH))
                c = zlib.compressobj(6, zlib.DEFLATED, -15)  # fresh dict
            chunks.append(body)
            seg_lens.append((len(body), hi - pos))
            pos = hi
            if pos >= n:
                break
        import struct as _struct

        payload = _struct.pack("<BBH", 1, 0, len(seg_lens)) + b"".join(
            _struct.pack("<II", c_, u_) for c_, u_ in seg_lens)
        extra = b"TS" + _struct.pack("<H", len(payload)) + payload
        hdr = (b"\x1f\x8b\x08\x04" + b"\x00\x00\x00\x00" + b"\x00\xff" +
               _struct.pack("<H", len(extra)) + extra)
        trailer = _struct.pack("<II", zlib.crc32(data) & 0xFFFFFFFF,
                               n % (1 << 32))
        return hdr + b"".join(chunks) + trailer
    if codec == "deflate":
        return zlib.compress(data, 6)
    raise ValueError(codec)


def parse_gz_segments(raw: bytes):
    """Parse our FEXTRA 'TS' segment table from a gzip blob.

    Returns (body_off, [(comp_len, uncomp_len), ...], crc32, isize) or None
    when the blob is not a single-member gzip with our table."""
    import struct as _struct

    if len(raw) < 20 or raw[:3] != b"\x1f\x8b\x08" or not (raw[3] & 0x04):
        return None
    flg = raw[3]
    if flg & ~0x04:  # any flag other than FEXTRA shifts fields we don't walk
        return None
    xlen = int.from_bytes(raw[10:12], "little")
    extra = raw[12:12 + xlen]
    body_off = 12 + xlen
    pos = 0
    while pos + 4 <= len(extra):
        si, ln = extra[pos:pos + 2], int.from_bytes(extra[pos + 2:pos + 4],
                                                    "little")
        sub = extra[pos + 4:pos + 4 + ln]
        pos += 4 + ln
        if si != b"TS" or len(sub) < 4:
            continue
        ver, _, nseg = _struct.unpack("<BBH", sub[:4])
        if ver != 1 or len(sub) < 4 + 8 * nseg:
            return None
        segs = [_struct.unpack("<II", sub[4 + 8 * i:12 + 8 * i])
                for i in range(nseg)]
        crc = int.from_bytes(raw[-8:-4], "little")
        isize = int.from_bytes(raw[-4:], "little")
        if sum(c_ for c_, _ in segs) != len(raw) - body_off - 8:
            return None  # truncated/concatenated: not a clean single member
        return body_off, segs, crc, isize
    return None


def _gunzip_parallel(raw: bytes) -> Optional[bytes]:
    """Parallel segmented inflate of a single-member gzip blob. With our
    FEXTRA table the segment extents are exact; a table-less blob written
    with full-flush sync points falls back to the marker scan. Returns None
    when neither fast path applies (multi-member, other header fields,
    marker false positive, CRC mismatch)."""
    meta = parse_gz_segments(raw)
    if meta is not None:
        body_off, seg_lens, crc_want, isize = meta
        segs = []
        pos = body_off
        for c, _u in seg_lens:
            segs.append(raw[pos:pos + c])
            pos += c
    else:
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


def parse_gz_segments_file(path: str):
    """parse_gz_segments reading only the header and trailer of a file:
    returns (body_off, [(comp_len, uncomp_len), ...], crc32, isize) or None.
    The GPU reader uses this to size device buffers without touching the
    compressed body on the host (it DMAs straight from the page cache)."""
    import struct as _struct

    try:
        size = os.path.getsize(path)
        if size < 20:
            return None
        with open(path, "rb") as f:
            head = f.read(12)
            if head[:3] != b"\x1f\x8b\x08" or not (head[3] & 0x04):
                return None
            if head[3] & ~0x04:
                return None
            xlen = int.from_bytes(head[10:12], "little")
            extra = f.read(xlen)
            f.seek(size - 8)
            trailer = f.read(8)
    except OSError:
        return None
    body_off = 12 + xlen
    pos = 0
    while pos + 4 <= len(extra):
        si = extra[pos:pos + 2]
        ln = int.from_bytes(extra[pos + 2:pos + 4], "little")
        sub = extra[pos + 4:pos + 4 + ln]
        pos += 4 + ln
        if si != b"TS" or len(sub) < 4:
            continue
        ver, _, nseg = _struct.unpack("<BBH", sub[:4])
        if ver != 1 or len(sub) < 4 + 8 * nseg:
            return None
        segs = [_struct.unpack("<II", sub[4 + 8 * i:12 + 8 * i])
                for i in range(nseg)]
        if sum(c_ for c_, _ in segs) != size - body_off - 8:
            return None
        crc = int.from_bytes(trailer[:4], "little")
        isize = int.from_bytes(trailer[4:], "little")
        return body_off, segs, crc, isize
    return None


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
