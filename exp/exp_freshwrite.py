import os, sys, time, ctypes
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from spark_tfrecord_amd import _native
from spark_tfrecord_amd.engine import gpu as g

N = 215 << 20
img = torch.zeros(N, dtype=torch.uint8, device="cuda")
torch.cuda.synchronize()
buf = g.pinned_buffer("fw2", N)

def fresh(path):
    try: os.unlink(path)
    except FileNotFoundError: pass

def timeit(tag, fn, reps=3):
    ts = []
    for i in range(reps):
        p = f"/dev/shm/fw_{tag}_{i}.bin"
        fresh(p)
        t = time.perf_counter(); fn(p); ts.append(time.perf_counter()-t)
        fresh(p)
    print(f"{tag}: best {min(ts)*1000:.1f} ms  ({N/min(ts)/1e9:.1f} GB/s)")

def via_register(p):
    ptr, pinned = _native.file_mmap_pinned(p, N, True)
    g._multi_dma(ptr, img.data_ptr(), N, _native.gpu_memcpy_d2h, after_main=False)
    _native.file_mmap_drop(p)

def d2h_stage():
    buf[:N].copy_(img, non_blocking=True)
    torch.cuda.synchronize()

def via_pwrite(p):
    d2h_stage()
    fd = os.open(p, os.O_RDWR | os.O_CREAT, 0o644)
    _native.pwrite_parallel(fd, buf.data_ptr(), N, 0)
    os.close(fd)

def via_memcpy(p):
    d2h_stage()
    ptr = _native.mmap_plain(p, N, True)
    _native.memcpy_parallel(ptr, buf.data_ptr(), N)
    _native.munmap_plain(ptr, N)

def via_write_syscall(p):
    d2h_stage()
    with open(p, "wb") as f:
        f.write(memoryview(buf.numpy()[:N]))

timeit("register_dma", via_register)
timeit("pwrite_pool", via_pwrite)
timeit("mmap_memcpy", via_memcpy)
timeit("plain_write", via_write_syscall)
