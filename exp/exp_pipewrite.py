"""Diagnose the pipelined single-shard write on the GPU box.

Phases: (a) full write with pipelined chunking vs forced one-shot,
(b) microbench of mmap-offset copies into a fresh /dev/shm file from a
pinned D2H view vs a plain host array, at 1/2/8 threads, vs write().
"""
import os, sys, time, mmap
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import spark_tfrecord_amd as stf
from bench import make_batch
from spark_tfrecord_amd.arrow_interop import batch_to_table
from spark_tfrecord_amd.engine import gpu as g
from spark_tfrecord_amd.io import writer as W
from concurrent.futures import ThreadPoolExecutor

batch = make_batch(1_000_000, seed=6)
table = batch_to_table(batch)
out = "/dev/shm/pipew"

def timed(name, fn, reps=3):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    print(f"{name:34s} {(time.perf_counter()-t0)/reps*1000:8.2f} ms", flush=True)

timed("write pipelined (default)", lambda: stf.write_tfrecord(
    table, out, engine="gpu", mode="overwrite"))
W._PIPE_CHUNK_ROWS = 10**9
timed("write one-shot (forced)", lambda: stf.write_tfrecord(
    table, out, engine="gpu", mode="overwrite"))
W._PIPE_CHUNK_ROWS = 131072

# microbench: copy 215 MB into a fresh tmpfs file via mmap at offsets
img = g.encode_device(g.batch_to_device(
    __import__("spark_tfrecord_amd.arrow_interop", fromlist=["table_to_batch"]
               ).table_to_batch(table, batch.schema)), "Example")
pinned = g.device_to_pinned_view(img, tag="probe")
host = np.array(pinned)  # plain pageable copy
N = pinned.nbytes
PAGE = mmap.PAGESIZE
pool = ThreadPoolExecutor(16)

def mmcopy(src, nt):
    p = "/dev/shm/pw_mm"
    try: os.unlink(p)
    except OSError: pass
    fd = os.open(p, os.O_CREAT | os.O_RDWR, 0o644)
    os.ftruncate(fd, N)
    step = ((N + nt - 1) // nt) // PAGE * PAGE + PAGE
    def cp(i):
        lo = i * step; hi = min(N, lo + step)
        if hi <= lo: return
        m = mmap.mmap(fd, hi - lo, offset=lo)
        d = np.frombuffer(m, dtype=np.uint8)
        d[:] = src[lo:hi]
        del d; m.close()
    if nt == 1: cp(0)
    else: list(pool.map(cp, range(nt)))
    os.close(fd); os.unlink(p)

def wrcopy(src):
    p = "/dev/shm/pw_wr"
    with open(p, "wb") as f:
        f.write(src)
    os.unlink(p)

for name, src in (("pinned", pinned), ("host", host)):
    for nt in (1, 2, 8):
        t0 = time.perf_counter(); mmcopy(src, nt); dt = time.perf_counter()-t0
        print(f"mmap-copy src={name:6s} nt={nt}   {dt*1000:8.2f} ms {N/dt/1e9:6.2f} GB/s", flush=True)
    t0 = time.perf_counter(); wrcopy(src); dt = time.perf_counter()-t0
    print(f"write()   src={name:6s}        {dt*1000:8.2f} ms {N/dt/1e9:6.2f} GB/s", flush=True)

# memory-to-memory read speed of the pinned view
dst = np.empty(N, np.uint8)
for name, src in (("pinned", pinned), ("host", host)):
    t0 = time.perf_counter(); dst[:] = src; dt = time.perf_counter()-t0
    print(f"mem copy  src={name:6s}        {dt*1000:8.2f} ms {N/dt/1e9:6.2f} GB/s", flush=True)
