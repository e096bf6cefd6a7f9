#!/usr/bin/env python3
"""Flagship benchmark: 1M-row Example TFRecord read+write round-trip.

BASELINE.json config 2: "1M-row Example with Int64List/FloatList/BytesList
array features, read+write on 1 MI355X". The reference publishes no numbers
(BASELINE.md), so this is the self-measured headline: whole-job rows/sec
(write + read of the same rows counts once) aggregated over all ranks.

One step = encode R rows into a framed TFRecord file image on the GPU,
write it to storage, read it back, and decode it to device columns.
Weak scaling: each rank round-trips its own R rows.

Contract (driver): rank 0 prints exactly one JSON line; timing brackets are
barrier + torch.cuda.synchronize on both sides; MAX elapsed over ranks.
"""

import argparse
import json
import os
import sys
import tempfile
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def make_batch(rows: int, seed: int):
    """Synthetic 1M-row-Example-shaped data: Int64List[8], FloatList[16],
    BytesList[2] array features + an int64 id (~220 B/record on the wire)."""
    import spark_tfrecord_amd as stf
    from spark_tfrecord_amd.columnar import RecordBatch, WireColumn
    from spark_tfrecord_amd.schema import KIND_BYTES, KIND_FLOAT, KIND_INT64

    rng = np.random.default_rng(seed)
    schema = stf.StructType([
        stf.StructField("id", stf.LongType(), True),
        stf.StructField("ints", stf.ArrayType(stf.LongType()), True),
        stf.StructField("floats", stf.ArrayType(stf.FloatType()), True),
        stf.StructField("tokens", stf.ArrayType(stf.StringType()), True),
    ])
    R = rows
    ones = np.ones(R, np.uint8)

    def aro(k):
        return np.arange(0, (R + 1) * k, k, dtype=np.int64)

    id_col = WireColumn(KIND_INT64, False, ones, aro(1),
                        rng.integers(0, 2**62, R).astype(np.int64))
    ints = WireColumn(KIND_INT64, False, ones, aro(8),
                      rng.integers(0, 2**31, 8 * R).astype(np.int64))
    floats = WireColumn(KIND_FLOAT, False, ones, aro(16),
                        rng.random(16 * R).astype(np.float32))
    # two ~12-byte tokens per row
    tok = np.frombuffer(
        b"".join(f"tok{i % 997:06d}-{i % 89:02d}".encode() for i in range(2 * min(R, 4096))),
        np.uint8)
    reps = (2 * R + 2 * min(R, 4096) - 1) // (2 * min(R, 4096))
    tok_data = np.tile(tok, reps)[: 2 * R * 12]
    tokens = WireColumn(KIND_BYTES, False, ones, aro(2), tok_data,
                        elem_off=np.arange(0, (2 * R + 1) * 12, 12, dtype=np.int64))
    return RecordBatch(schema, [id_col, ints, floats, tokens], R)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--rows", type=int, default=1_000_000)
    ap.add_argument("--dir", type=str, default=None,
                    help="bench storage dir (default: /dev/shm tmpfs)")
    ap.add_argument("--engine", type=str, default="auto")
    ap.add_argument("--phases", action="store_true",
                    help="print per-phase timing breakdown (rank 0)")
    args = ap.parse_args()

    import torch

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank)

    dist = None
    if world_size > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        dist.init_process_group(backend="nccl" if use_cuda else "gloo")

    import spark_tfrecord_amd  # noqa: F401  (after device selection)
    from spark_tfrecord_amd.engine import cpu as cpu_engine
    from spark_tfrecord_amd import _native

    rows = args.rows if use_cuda else min(args.rows, 50_000)
    batch = make_batch(rows, seed=1234 + rank)

    base_dir = args.dir or ("/dev/shm" if os.path.isdir("/dev/shm") else
                            tempfile.gettempdir())
    work_dir = os.path.join(base_dir, f"tfrec_bench_r{rank}")
    os.makedirs(work_dir, exist_ok=True)
    fpath = os.path.join(work_dir, "bench.tfrecord")

    if use_cuda and args.engine in ("auto", "gpu"):
        from spark_tfrecord_amd.engine import gpu as gpu_engine

        dev_batch = gpu_engine.batch_to_device(batch)
        phase_t = {}

        def timed(name, fn):
            if not args.phases:
                return fn()
            torch.cuda.synchronize()
            t = time.perf_counter()
            r = fn()
            torch.cuda.synchronize()
            phase_t[name] = phase_t.get(name, 0.0) + time.perf_counter() - t
            return r

        def step():
            # write: sliced emit kernels overlapped with D2H DMA into the
            # mapped file; read: sliced H2D overlapped with the frame scan,
            # then CRC-verified decode to device columns
            timed("write", lambda: gpu_engine.write_batch_to_file(
                dev_batch, fpath, "Example"))
            out = timed("read+decode", lambda: gpu_engine.read_file_to_batch_pipelined(
                fpath, batch.schema, "Example", verify_crc=True))
            return out
        engine_name = "gpu"
    else:
        def step():
            img = cpu_engine.encode_batch(batch, "Example")
            with open(fpath, "wb") as f:
                f.write(img)
            data = np.fromfile(fpath, np.uint8)
            return cpu_engine.decode_buffer(data, batch.schema, "Example")
        engine_name = "cpu"

    # one correctness probe before timing: decoded ids must match
    out = step()
    if engine_name == "gpu":
        from spark_tfrecord_amd.engine import gpu as gpu_engine
        got = out.columns[0].values.cpu().numpy()
    else:
        got = np.asarray(out.columns[0].values)
    assert np.array_equal(got, np.asarray(batch.columns[0].values)), \
        "round-trip mismatch"
    file_bytes = os.path.getsize(fpath)

    def sync():
        if use_cuda:
            torch.cuda.synchronize()
        if dist is not None:
            dist.barrier()

    for _ in range(args.warmup):
        step()
    sync()
    if args.phases and engine_name == "gpu":
        phase_t.clear()  # report timed steps only (drop warmup one-offs)
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    sync()
    elapsed = time.perf_counter() - t0

    if dist is not None:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cuda" if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    n_gpus = world_size if use_cuda else args.gpus
    total_rows = rows * args.steps * world_size
    rows_per_sec = total_rows / elapsed
    mb_per_sec = file_bytes * args.steps * world_size / elapsed / 1e6

    if rank == 0 and args.phases and engine_name == "gpu":
        breakdown = {k: f"{v / args.steps * 1000:.1f}ms"
                     for k, v in phase_t.items()}
        print(f"# phase breakdown (per timed step): {breakdown}",
              file=sys.stderr)

    if rank == 0:
        print(json.dumps({
            "metric": "rows/sec read+write (1M-row Example round-trip)",
            "value": rows_per_sec,
            "unit": "rows/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "wire(int64/float32/bytes)",
            "data": "synthetic",
            "config": {
                "model": "Example{id:int64, ints:Int64List[8], floats:FloatList[16], tokens:BytesList[2]}",
                "global_batch": rows * world_size,
                "seq_len": None,
                "parallelism": f"dp{world_size}",
                "rows_per_rank": rows,
                "file_mb": round(file_bytes / 1e6, 2),
                "mb_per_sec_roundtrip": round(mb_per_sec, 1),
                "engine": engine_name,
                "storage": base_dir,
                "crc_verify": True,
            },
        }))

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
