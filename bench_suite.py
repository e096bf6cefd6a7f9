#!/usr/bin/env python3
"""Benchmark suite for the non-flagship BASELINE.json configs.

bench.py measures config 2 (the headline 1M-row Example round-trip; the
driver runs it). This suite covers the rest, each printing one JSON line:

  plumbing        config 1: 1k-row scalar Example round-trip, CPU engine
                  (the reference's Spark local[2] analog; 2 shards)
  partitionby     config 3: Example partitionBy("date") write; with
                  WORLD_SIZE>1 the rows are exchanged with the RCCL
                  all-to-all so each date has one writer rank
  infer           config 4: SequenceExample (FeatureList of FloatList)
                  schema inference; with WORLD_SIZE>1 the per-feature
                  lattice codes are max-all-reduced over RCCL
  gzip_bytearray  config 5: gzip-compressed ByteArray read, shard staged
                  through HBM on the GPU engine

Single process:  python bench_suite.py all --rows 100000
Multi GPU:       torchrun --nproc-per-node 8 bench_suite.py partitionby
"""

import argparse
import json
import os
import shutil
import sys
import tempfile
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def _env_world():
    return (int(os.environ.get("RANK", "0")),
            int(os.environ.get("WORLD_SIZE", "1")))


def _emit(rank, metric, value, unit, config, elapsed, n_gpus):
    if rank == 0:
        print(json.dumps({
            "metric": metric, "value": value, "unit": unit,
            "n_gpus": n_gpus, "elapsed_s": round(elapsed, 4),
            "higher_is_better": True, "data": "synthetic", "config": config,
        }), flush=True)


def _workdir(tag, rank):
    base = "/dev/shm" if os.path.isdir("/dev/shm") else tempfile.gettempdir()
    d = os.path.join(base, f"tfrec_suite_{tag}_r{rank}")
    shutil.rmtree(d, ignore_errors=True)
    os.makedirs(d, exist_ok=True)
    return d


def bench_plumbing(args):
    """Config 1: 1k-row scalar-only DataFrame round-trip on CPU, 2 shards."""
    import spark_tfrecord_amd as stf

    rng = np.random.default_rng(0)
    n = 1000
    data = {"lng": rng.integers(0, 2**40, n),
            "flt": rng.random(n).astype(np.float32),
            "s": [f"row-{i}" for i in range(n)]}
    d = _workdir("plumbing", 0)
    out = os.path.join(d, "t")
    reps = args.reps
    # one untimed warmup round-trip (module/thread-pool/arrow first-call costs)
    stf.write_tfrecord(data, out, engine="cpu", mode="overwrite", num_shards=2)
    stf.read_tfrecord(out, engine="cpu")
    t0 = time.perf_counter()
    for r in range(reps):
        stf.write_tfrecord(data, out, engine="cpu", mode="overwrite",
                           num_shards=2)
        df = stf.read_tfrecord(out, engine="cpu")
        assert df.count() == n
    el = time.perf_counter() - t0
    _emit(0, "rows/sec round-trip (1k-row scalar CPU plumbing)",
          n * reps / el, "rows/s",
          {"rows": n, "shards": 2, "engine": "cpu", "reps": reps}, el, 0)


def bench_partitionby(args):
    """Config 3: partitionBy("date") write; all-to-all shuffle when world>1."""
    import torch

    from spark_tfrecord_amd.parallel import dist as D

    rank, world = _env_world()
    use_cuda = torch.cuda.is_available()
    if world > 1:
        D.init_distributed()
    import pyarrow as pa

    rows = args.rows // world
    rng = np.random.default_rng(100 + rank)
    # build the arrow table ONCE: the timed loop measures the engine, not
    # python->arrow input conversion
    data = pa.table({
        "date": pa.array([f"2026-09-{d:02d}" for d in
                          rng.integers(1, 11, rows)]),
        "uid": pa.array(rng.integers(0, 2**62, rows)),
        "score": pa.array(rng.random(rows).astype(np.float32)),
        "feats": pa.array(list(rng.random((rows, 8)).astype(np.float32))),
    })
    # shared dir: ranks write distinct part files; rank 0 alone prepares it
    if world == 1 or rank == 0:
        d = _workdir("partby", 0)
    else:
        base = "/dev/shm" if os.path.isdir("/dev/shm") else tempfile.gettempdir()
        d = os.path.join(base, "tfrec_suite_partby_r0")
    out = os.path.join(d, "t")
    eng = "gpu" if use_cuda else "cpu"

    def sync():
        if use_cuda:
            torch.cuda.synchronize()
        if world > 1:
            import torch.distributed as td
            td.barrier()

    sync()  # setup rmtree (rank 0) must complete before anyone writes

    def one_rep():
        if world > 1:
            D.write_tfrecord_distributed(data, out, partition_by=["date"],
                                         mode="overwrite", engine=eng)
        else:
            import spark_tfrecord_amd as stf
            stf.write_tfrecord(data, out, partition_by=["date"],
                               mode="overwrite", engine=eng)

    one_rep()  # untimed warmup
    sync()
    t0 = time.perf_counter()
    for r in range(args.reps):
        one_rep()
    sync()
    el = time.perf_counter() - t0
    total = rows * world * args.reps
    _emit(rank, "rows/sec partitionBy write (all-to-all shuffle)",
          total / el, "rows/s",
          {"rows_total": rows * world, "partitions": 10, "engine": eng,
           "parallelism": f"dp{world}", "reps": args.reps}, el,
          world if use_cuda else 0)


def bench_infer(args):
    """Config 4: SequenceExample schema inference (+ all-reduce when world>1)."""
    import torch

    import spark_tfrecord_amd as stf
    from spark_tfrecord_amd.io.reader import infer_schema_of_paths
    from spark_tfrecord_amd.parallel import dist as D

    rank, world = _env_world()
    use_cuda = torch.cuda.is_available()
    if world > 1:
        D.init_distributed()

    def sync():
        if use_cuda:
            torch.cuda.synchronize()
        if world > 1:
            import torch.distributed as td
            td.barrier()

    # ONE shared dataset (the distributed contract: every rank passes the
    # SAME file list; rank r scans records r::world of the chosen file and
    # the lattice codes are max-all-reduced)
    rows = args.rows
    d = _workdir("infer", 0) if (world == 1 or rank == 0) else None
    if d is None:
        base = "/dev/shm" if os.path.isdir("/dev/shm") else tempfile.gettempdir()
        d = os.path.join(base, "tfrec_suite_infer_r0")
    out = os.path.join(d, "t")
    if rank == 0:
        rng = np.random.default_rng(7)
        # ragged 2-D: FeatureList of FloatList
        rag = [[list(rng.random(rng.integers(1, 6)).astype(float))
                for _ in range(int(rng.integers(1, 5)))] for _ in range(rows)]
        data = {"sid": np.arange(rows, dtype=np.int64), "rag": rag}
        schema = stf.StructType([
            stf.StructField("sid", stf.LongType(), True),
            stf.StructField("rag",
                            stf.ArrayType(stf.ArrayType(stf.FloatType())), True),
        ])
        stf.write_tfrecord(data, out, record_type="SequenceExample",
                           schema=schema,
                           engine="gpu" if use_cuda else "cpu",
                           mode="overwrite")
    sync()
    files = [os.path.join(out, f) for f in sorted(os.listdir(out))
             if not f.startswith("_")]

    def one_rep():
        if world > 1:
            return D.infer_schema_distributed(files, "SequenceExample")
        return infer_schema_of_paths(files, "SequenceExample",
                                     "gpu" if use_cuda else "cpu")

    s = one_rep()  # untimed warmup
    sync()
    t0 = time.perf_counter()
    for _ in range(args.reps):
        s = one_rep()
    sync()
    el = time.perf_counter() - t0
    assert "rag" in [f.name for f in s.fields]
    total = rows * args.reps  # shared file: `rows` records scanned per rep
    _emit(rank, "records/sec schema inference (SequenceExample lattice)",
          total / el, "records/s",
          {"rows_total": rows, "parallelism": f"dp{world}",
           "engine": "gpu" if use_cuda else "cpu", "reps": args.reps}, el,
          world if use_cuda else 0)


def bench_gzip_bytearray(args):
    """Config 5: gzip ByteArray read, shard staged through HBM."""
    import torch

    import spark_tfrecord_amd as stf

    from spark_tfrecord_amd.parallel import dist as D

    rank, world = _env_world()
    use_cuda = torch.cuda.is_available()
    if world > 1:
        D.init_distributed()

    def sync():
        if use_cuda:
            torch.cuda.synchronize()
        if world > 1:
            import torch.distributed as td
            td.barrier()

    rows = args.rows // max(world, 1)
    rng = np.random.default_rng(9 + rank)
    payloads = [rng.bytes(200) for _ in range(rows)]
    import pyarrow as pa
    table = pa.table({"byteArray": pa.array(payloads, type=pa.large_binary())})
    d = _workdir("gzba", rank)
    out = os.path.join(d, "t")
    # many shards, like one Spark task per file in the reference: gzip is
    # sequential PER file, so shards are the parallelism axis on read
    stf.write_tfrecord(table, out, record_type="ByteArray", codec="gzip",
                       mode="overwrite", engine="cpu", num_shards=32)
    nbytes = sum(os.path.getsize(os.path.join(out, f))
                 for f in os.listdir(out) if not f.startswith("_"))
    eng = "gpu" if use_cuda else "cpu"
    stf.read_tfrecord(out, record_type="ByteArray", engine=eng)  # warmup
    sync()
    t0 = time.perf_counter()
    for _ in range(args.reps):
        df = stf.read_tfrecord(out, record_type="ByteArray", engine=eng)
        assert df.count() == rows
    sync()
    el = time.perf_counter() - t0
    _emit(rank, "rows/sec gzip ByteArray read",
          rows * world * args.reps / el, "rows/s",
          {"rows_total": rows * world, "gz_bytes_per_rank": nbytes,
           "engine": eng, "parallelism": f"dp{world}", "reps": args.reps}, el,
          world if use_cuda else 0)


BENCHES = {"plumbing": bench_plumbing, "partitionby": bench_partitionby,
           "infer": bench_infer, "gzip_bytearray": bench_gzip_bytearray}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("which", choices=list(BENCHES) + ["all"])
    ap.add_argument("--rows", type=int, default=1_000_000)
    ap.add_argument("--reps", type=int, default=3)
    args = ap.parse_args()
    names = list(BENCHES) if args.which == "all" else [args.which]
    for n in names:
        BENCHES[n](args)


if __name__ == "__main__":
    main()
