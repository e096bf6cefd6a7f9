"""Probe D2H/H2D bandwidth: hipHostMalloc vs registered-mmap dst, 1 vs 4 streams."""
import os, sys, time, mmap
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from spark_tfrecord_amd import _native

N = 256 << 20
dev = torch.empty(N, dtype=torch.uint8, device="cuda")
dev.fill_(7)
torch.cuda.synchronize()

def timeit(fn, reps=5):
    fn(); torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / reps

streams = [torch.cuda.Stream() for _ in range(4)]
main = torch.cuda.current_stream()

def copy(dst_ptr, src_ptr, k, fn):
    span = (N + k - 1) // k
    for i in range(k):
        o = i * span; m = min(span, N - o)
        s = streams[i]; s.wait_stream(main)
        fn(dst_ptr + o, src_ptr + o, m, s.cuda_stream)
    for i in range(k): main.wait_stream(streams[i])

# pinned (hipHostMalloc via torch)
pin = torch.empty(N, dtype=torch.uint8, pin_memory=True)
for k in (1, 2, 4):
    t = timeit(lambda: copy(pin.data_ptr(), dev.data_ptr(), k, _native.gpu_memcpy_d2h))
    print(f"D2H pinned   k={k}: {N/t/1e9:.1f} GB/s")
for k in (1, 2, 4):
    t = timeit(lambda: copy(dev.data_ptr(), pin.data_ptr(), k, _native.gpu_memcpy_h2d))
    print(f"H2D pinned   k={k}: {N/t/1e9:.1f} GB/s")

# registered mmap on /dev/shm
path = "/dev/shm/dma_probe.bin"
ptr, pinned = _native.file_mmap_pinned(path, N, True)
print("mmap pinned:", pinned)
if pinned:
    for k in (1, 2, 4):
        t = timeit(lambda: copy(ptr, dev.data_ptr(), k, _native.gpu_memcpy_d2h))
        print(f"D2H regmmap  k={k}: {N/t/1e9:.1f} GB/s")
    for k in (1, 2, 4):
        t = timeit(lambda: copy(dev.data_ptr(), ptr, k, _native.gpu_memcpy_h2d))
        print(f"H2D regmmap  k={k}: {N/t/1e9:.1f} GB/s")
