#!/usr/bin/env python3
"""Micro A/B experiments for the host<->device<->storage legs of the pipeline
(run on the GPU box; results inform engine/gpu.py buffering strategy)."""

import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

N = 215 * 1024 * 1024  # ~bench file size
REPS = 8


def bench(name, fn, reps=REPS):
    fn()  # warm
    torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t) / reps
    print(f"{name:40s} {dt * 1000:8.2f} ms   {N / dt / 1e9:7.2f} GB/s", flush=True)
    return dt


def main():
    dev = torch.empty(N, dtype=torch.uint8, device="cuda")
    dev.random_(0, 255)
    pinned = torch.empty(N, dtype=torch.uint8, pin_memory=True)
    pageable = torch.empty(N, dtype=torch.uint8)

    print("== D2H ==")
    bench("d2h pageable (.cpu())", lambda: dev.cpu())
    bench("d2h pinned copy_", lambda: pinned.copy_(dev, non_blocking=True))

    print("== H2D ==")
    host_np = np.random.randint(0, 255, N, dtype=np.uint8)
    bench("h2d pageable as_tensor.to", lambda: torch.as_tensor(host_np).cuda())
    pinned_np = pinned.numpy()
    def h2d_pinned():
        dev.copy_(pinned, non_blocking=True)
    bench("h2d pinned copy_", h2d_pinned)

    print("== file write (tmpfs /dev/shm) ==")
    path = "/dev/shm/exp_io.bin"
    arr = pinned.numpy()

    def w_tofile_trunc():
        with open(path, "wb") as f:
            arr.tofile(f)
    bench("write tofile (trunc)", w_tofile_trunc)

    def w_inplace():
        with open(path, "r+b") as f:
            f.write(memoryview(arr))
    bench("write r+b in place", w_inplace)

    fd = os.open(path, os.O_WRONLY)
    def w_pwrite():
        os.pwrite(fd, memoryview(arr), 0)
    bench("write pwrite in place", w_pwrite)
    os.close(fd)

    print("== file read ==")
    bench("read np.fromfile", lambda: np.fromfile(path, np.uint8))

    buf = np.empty(N, np.uint8)
    def r_into():
        with open(path, "rb") as f:
            f.readinto(memoryview(buf))
    bench("read readinto pageable", r_into)

    def r_into_pinned():
        with open(path, "rb") as f:
            f.readinto(memoryview(pinned_np))
    bench("read readinto pinned", r_into_pinned)

    print("== host frame header scan (reference: current decode path) ==")
    import spark_tfrecord_amd as stf
    from bench import make_batch
    from spark_tfrecord_amd.engine import cpu as cpu_engine
    from spark_tfrecord_amd import _native

    batch = make_batch(1_000_000, seed=3)
    img = cpu_engine.encode_batch(batch, "Example")
    data = np.frombuffer(img, np.uint8)
    t = time.perf_counter()
    for _ in range(3):
        off, lens = _native.scan_frame_headers(data)
    print(f"scan_frame_headers 1M rec {(time.perf_counter() - t) / 3 * 1000:.2f} ms",
          flush=True)


if __name__ == "__main__":
    main()
