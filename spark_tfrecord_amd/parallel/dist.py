"""Distributed execution: one process per GPU over torch.distributed.

MI355X-first design: the reference delegates distribution to
Spark (tasks, shuffle, tree-aggregate); here the equivalents are explicit
collectives over RCCL/xGMI (backend "nccl" on ROCm) with a gloo fallback for
CPU-only plumbing:

  - read/write shard fan-out: files (read) and part-file shards (write) are
    owned round-robin by rank — the analog of one Spark task per file
    (isSplitable=false, DefaultSource.scala:26-29);
  - schema inference: each rank scans a record-range slice of the first
    non-empty file, then codes are max-all-reduced (the lattice merge is
    commutative/associative, TensorFlowInferSchema.scala:120-127);
  - partitionBy shuffle: rows are exchanged with an all-to-all so each
    partition value has exactly one writer rank. Records are exchanged
    ALREADY ENCODED: TFRecord frames are concatenable, so the source rank
    encodes once (on its GPU) and receivers just concatenate and write —
    xGMI moves wire bytes, never re-serialized rows.
"""

from __future__ import annotations

import hashlib
import os
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch
import torch.distributed as dist

from .. import _native
from ..engine import resolve_engine
from ..infer import byte_array_schema, infer_codes_from_buffer, schema_from_codes
from ..io import paths as P
from ..schema import StructType, validate_schema_for_record_type

__all__ = ["init_distributed", "shard_files", "infer_schema_distributed",
           "write_tfrecord_distributed", "read_tfrecord_distributed"]


def init_distributed(backend: Optional[str] = None):
    """Initialize from torchrun env vars; idempotent. Returns (rank, world)."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    dist.init_process_group(backend=backend)
    return dist.get_rank(), dist.get_world_size()


def _world() -> Tuple[int, int]:
    if dist.is_available() and dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    return 0, 1


def _comm_device() -> torch.device:
    if dist.get_backend() == "nccl":
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def shard_files(files: Sequence[str], rank: int, world: int) -> List[str]:
    """Round-robin file ownership (one whole file per task, like the
    reference's unsplittable reads)."""
    return [f for i, f in enumerate(files) if i % world == rank]


# ---------------------------------------------------------------------------
# Distributed schema inference
# ---------------------------------------------------------------------------

def infer_schema_distributed(files: List[str], record_type: str) -> StructType:
    """All ranks return the same schema.

    Mirrors the reference's rule (DefaultSource.scala:36-38): the FIRST
    non-empty file decides, scanned fully — here its records are split across
    ranks and the per-feature lattice codes are max-all-reduced over
    RCCL/xGMI (gloo on CPU).
    """
    rank, world = _world()
    if record_type == "ByteArray":
        return byte_array_schema()
    if world == 1:
        from ..io.reader import infer_schema_of_paths
        return infer_schema_of_paths(files, record_type)

    dev = _comm_device()
    # rank 0 finds the first non-empty file; everyone agrees via broadcast
    chosen = -1
    if rank == 0:
        for i, f in enumerate(files):
            data = np.frombuffer(P.decompress_file(f), np.uint8)
            if data.size:
                chosen = i
                break
    t = torch.tensor([chosen], dtype=torch.int64, device=dev)
    dist.broadcast(t, src=0)
    chosen = int(t.item())
    if chosen < 0:
        raise ValueError("Could not infer schema: no non-empty TFRecord files found")

    data = np.frombuffer(P.decompress_file(files[chosen]), np.uint8)
    off, lens = _native.scan_frames(data, False)
    # each rank scans an interleaved slice of the records — on its GPU via
    # the hash-table lattice kernel when one is available
    my_off = off[rank::world]
    my_len = lens[rank::world]
    if len(my_off) == 0:
        codes: Dict[str, int] = {}
    elif torch.cuda.is_available():
        from ..engine import gpu as gpu_engine

        dev_data = torch.as_tensor(np.ascontiguousarray(data)).cuda()
        codes = gpu_engine.infer_codes_device(
            dev_data, torch.as_tensor(np.ascontiguousarray(my_off)).cuda(),
            torch.as_tensor(np.ascontiguousarray(my_len)).cuda(), record_type)
    else:
        codes = infer_codes_from_buffer(data, my_off, my_len, record_type)

    # align feature names across ranks, then max-all-reduce the code vector
    gathered: List[Dict[str, int]] = [None] * world
    dist.all_gather_object(gathered, {k: 0 for k in codes})
    names = sorted(set().union(*[set(g) for g in gathered]))
    vec = torch.zeros(max(len(names), 1), dtype=torch.int64, device=dev)
    for i, n in enumerate(names):
        vec[i] = codes.get(n, 0)
    dist.all_reduce(vec, op=dist.ReduceOp.MAX)
    merged = {n: int(vec[i]) for i, n in enumerate(names)}
    return schema_from_codes(merged)


# ---------------------------------------------------------------------------
# Distributed write (+ partitionBy all-to-all)
# ---------------------------------------------------------------------------

def _owner_of(partition_key: Tuple, world: int) -> int:
    h = hashlib.blake2s(repr(partition_key).encode(), digest_size=8).digest()
    return int.from_bytes(h, "little") % world


def _blob_len(b) -> int:
    return b.numel() if isinstance(b, torch.Tensor) else len(b)


def _all_to_all_blobs(send: List, dev: torch.device) -> List:
    """Exchange one blob per peer. Blobs may be bytes or torch.uint8
    tensors. nccl: a true all-to-all of DEVICE tensors over xGMI — encoded
    partition bytes move GPU-to-GPU and come back as device tensors (the
    receiving rank DMAs them straight into its partition file, no host
    round-trip). gloo (CPU tests): emulated with all_gather per
    destination, bytes in/out."""
    rank, world = _world()
    if dist.get_backend() == "nccl":
        in_sizes = torch.tensor([_blob_len(b) for b in send], dtype=torch.int64,
                                device=dev)
        out_sizes = torch.empty(world, dtype=torch.int64, device=dev)
        dist.all_to_all_single(out_sizes, in_sizes)
        send_t = [b.to(dev).contiguous() if isinstance(b, torch.Tensor) else
                  (torch.frombuffer(bytearray(b), dtype=torch.uint8).to(dev)
                   if len(b) else torch.zeros(0, dtype=torch.uint8, device=dev))
                  for b in send]
        recv_t = [torch.empty(int(s), dtype=torch.uint8, device=dev)
                  for s in out_sizes]
        dist.all_to_all(recv_t, send_t)
        return recv_t
    send = [bytes(b.cpu().numpy().tobytes()) if isinstance(b, torch.Tensor)
            else b for b in send]
    # gloo fallback: for each destination d, gather everyone's blob-to-d
    out: List[bytes] = [b""] * world
    for d in range(world):
        blob = torch.frombuffer(bytearray(send[d]), dtype=torch.uint8) \
            if len(send[d]) else torch.zeros(0, dtype=torch.uint8)
        sizes = [torch.zeros(1, dtype=torch.int64) for _ in range(world)]
        dist.all_gather(sizes, torch.tensor([blob.numel()], dtype=torch.int64))
        mx = int(max(s.item() for s in sizes))
        padded = torch.zeros(mx, dtype=torch.uint8)
        padded[: blob.numel()] = blob
        bufs = [torch.empty(mx, dtype=torch.uint8) for _ in range(world)]
        dist.all_gather(bufs, padded)
        if d == rank:
            out = [bytes(bufs[s][: int(sizes[s].item())].numpy().tobytes())
                   for s in range(world)]
    return out


def write_tfrecord_distributed(data, path: str, record_type: str = "Example",
                               codec: Optional[str] = None,
                               mode: str = "errorifexists",
                               partition_by: Optional[Sequence[str]] = None,
                               schema: Optional[StructType] = None,
                               engine: str = "auto") -> None:
    """Each rank writes its own shard of `data` (rank-local rows).

    Without partitionBy: rank r writes part file r. With partitionBy: rows
    are grouped by partition value, ENCODED on the owning source rank, then
    the wire bytes are exchanged all-to-all so each partition directory is
    written by exactly one rank (deterministic hash ownership).
    """
    import pyarrow as pa

    from ..arrow_interop import schema_from_arrow, table_to_batch
    from ..engine import cpu as cpu_engine
    from ..io.writer import _partition_dir_value, normalize_input

    rank, world = _world()
    if world == 1:
        from ..io.writer import write_tfrecord
        return write_tfrecord(data, path, record_type=record_type, codec=codec,
                              mode=mode, partition_by=partition_by, schema=schema,
                              engine=engine)

    dev = _comm_device()
    codec = P.normalize_codec(codec)
    table = normalize_input(data, schema)
    if schema is None:
        schema = schema_from_arrow(table.schema)
    validate_schema_for_record_type(schema, record_type)
    eng = resolve_engine(engine)

    # save-mode coordination: rank 0 prepares, all wait, everyone honors skip
    proceed = 1
    if rank == 0:
        proceed = 1 if P.apply_save_mode(path, mode) else 0
    t = torch.tensor([proceed], dtype=torch.int64, device=dev)
    dist.broadcast(t, src=0)
    if int(t.item()) == 0:
        return
    dist.barrier()

    job_id = f"{rank:03d}" + hashlib.blake2s(
        f"{path}:{rank}".encode(), digest_size=4).hexdigest()

    def encode(tbl, sch) -> bytes:
        batch = table_to_batch(tbl, sch)
        if eng == "gpu":
            from ..engine import gpu as gpu_engine
            return gpu_engine.encode_batch_from_cpu(batch, record_type)
        return cpu_engine.encode_batch(batch, record_type)

    if not partition_by:
        raw = encode(table, schema)
        fname = P.part_file_name(rank, codec, job_id)
        P.write_file_atomic(P.compress_bytes(raw, codec),
                            os.path.join(path, fname))
    else:
        import numpy as np

        from ..io.writer import _factorize_partitions

        data_cols = [c for c in table.column_names if c not in set(partition_by)]
        data_schema = StructType([f for f in schema.fields
                                  if f.name in set(data_cols)])
        codes, combos = _factorize_partitions(table, partition_by)
        stripped = table.select(data_cols)
        # per-partition wire blobs: on the GPU every row is encoded ONCE and
        # the partition split is a framed-record gather in HBM — the blobs
        # stay device tensors all the way through the RCCL all-to-all
        part_blobs: Dict[int, object] = {}
        if eng == "gpu" and record_type != "ByteArray" and table.num_rows > 0:
            from ..engine import gpu as gpu_engine

            batch = gpu_engine.batch_to_device(
                table_to_batch(stripped, data_schema))
            img, ranges = gpu_engine.encode_partitions_device(
                batch, codes, len(combos), record_type)
            for p, lo, hi in ranges:
                part_blobs[p] = img[lo:hi]
        else:
            for p in range(len(combos)):
                idxs = np.nonzero(codes == p)[0]
                if idxs.size == 0:
                    continue
                sub = stripped.take(pa.array(idxs, type=pa.int64()))
                part_blobs[p] = encode(sub, data_schema)
        send_parts: List[List[Tuple[tuple, object]]] = [[] for _ in range(world)]
        for p in sorted(part_blobs):
            send_parts[_owner_of(combos[p], world)].append(
                (combos[p], part_blobs[p]))
        # header: list[(combo, length)] exchanged as objects (tiny)
        headers: List[List[Tuple[tuple, int]]] = [
            [(c, _blob_len(b)) for c, b in parts] for parts in send_parts]
        gathered_headers: List[List[List[Tuple[tuple, int]]]] = [None] * world
        dist.all_gather_object(gathered_headers, headers)

        def _concat(parts):
            blobs = [b for _, b in parts]
            if blobs and isinstance(blobs[0], torch.Tensor):
                return (torch.cat(blobs) if len(blobs) > 1
                        else blobs[0])
            return b"".join(blobs)

        recv_blobs = _all_to_all_blobs([_concat(p) for p in send_parts], dev)
        # stitch: per incoming rank, split by its header for me (slicing a
        # device tensor is a view — no copy)
        mine: Dict[tuple, List] = {}
        for src in range(world):
            hdr = gathered_headers[src][rank]
            blob = recv_blobs[src]
            pos = 0
            for combo, ln in hdr:
                mine.setdefault(combo, []).append(blob[pos:pos + ln])
                pos += ln
        # device blobs: concatenate into ONE image, one pinned D2H, then
        # write the per-partition files from view slices on a thread pool
        # (per-file mmap registration on fresh inodes costs more than the
        # bytes' DMA — see io/writer.py partition path)
        dev_jobs: List[Tuple[tuple, int, int]] = []
        host_jobs: List[Tuple[tuple, bytes]] = []
        dev_parts: List[torch.Tensor] = []
        pos = 0
        for combo, blobs in sorted(mine.items(), key=lambda kv: str(kv[0])):
            if blobs and isinstance(blobs[0], torch.Tensor):
                raw_t = (torch.cat(blobs) if len(blobs) > 1
                         else blobs[0]).contiguous()
                dev_parts.append(raw_t)
                dev_jobs.append((combo, pos, pos + raw_t.numel()))
                pos += raw_t.numel()
            else:
                host_jobs.append((combo, b"".join(blobs)))
        view = None
        if dev_parts:
            from ..engine import gpu as gpu_engine

            img = torch.cat(dev_parts) if len(dev_parts) > 1 else dev_parts[0]
            view = gpu_engine.device_to_pinned_view(img, tag="dist_d2h")

        def _write_combo(combo, payload):
            sub_dir = os.path.join(
                path, *(f"{c}={_partition_dir_value(v)}"
                        for c, v in zip(partition_by, combo)))
            os.makedirs(sub_dir, exist_ok=True)
            fpath = os.path.join(sub_dir, P.part_file_name(rank, codec, job_id))
            P.write_file_atomic(
                payload if codec is None else P.compress_bytes(
                    payload if isinstance(payload, bytes)
                    else payload.tobytes(), codec), fpath)

        njobs = len(dev_jobs) + len(host_jobs)
        if njobs:
            ex = P.shared_pool()
            futs = [ex.submit(_write_combo, c, view[lo:hi])
                    for c, lo, hi in dev_jobs]
            futs += [ex.submit(_write_combo, c, b) for c, b in host_jobs]
            for f in futs:
                f.result()

    dist.barrier()
    if rank == 0:
        P.write_success_marker(path)
    dist.barrier()


def read_tfrecord_distributed(path: str, schema: Optional[StructType] = None,
                              record_type: str = "Example",
                              engine: str = "auto", verify_crc: bool = True):
    """Rank-local shard of the dataset as a DataFrame (files round-robin)."""
    rank, world = _world()
    if world == 1:
        from ..io.reader import read_tfrecord
        return read_tfrecord(path, schema=schema, record_type=record_type,
                             engine=engine, verify_crc=verify_crc)
    files = P.list_data_files(path)
    if not files:
        raise FileNotFoundError(f"No TFRecord files found under {path}")
    if schema is None:
        schema = infer_schema_distributed(files, record_type)
    my_files = shard_files(files, rank, world)
    from ..api import DataFrame
    from ..arrow_interop import schema_to_arrow
    from ..io.reader import read_tfrecord

    if not my_files:
        import pyarrow as pa
        empty = pa.table({f.name: [] for f in schema.fields},
                         schema=schema_to_arrow(schema))
        return DataFrame(empty, schema)
    root = path if os.path.isdir(path) else os.path.dirname(path)
    # ONE call over the rank's whole file list: the GPU reader batches
    # consecutive uncompressed files into a single scan+decode pipeline;
    # partition-column discovery stays relative to the dataset root
    df = read_tfrecord(my_files, schema=schema, record_type=record_type,
                       engine=engine, verify_crc=verify_crc, base_dir=root)
    return df
