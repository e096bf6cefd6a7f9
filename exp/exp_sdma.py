"""Does H2D/D2H to registered-mmap vs torch-pinned memory use SDMA or blit kernels?"""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from spark_tfrecord_amd import _native

N = 256 << 20
dev = torch.zeros(N, dtype=torch.uint8, device="cuda")
pin = torch.empty(N, dtype=torch.uint8, pin_memory=True)
ptr, pinned = _native.file_mmap_pinned("/dev/shm/sdma_probe.bin", N, True)
assert pinned
s = torch.cuda.Stream()
torch.cuda.synchronize()

def run(tag, dst, src, fn):
    t = time.perf_counter()
    for _ in range(3):
        fn(dst, src, N, s.cuda_stream)
    s.synchronize()
    print(f"{tag}: {3*N/(time.perf_counter()-t)/1e9:.1f} GB/s")

run("D2H->pinned ", pin.data_ptr(), dev.data_ptr(), _native.gpu_memcpy_d2h)
run("D2H->regmmap", ptr, dev.data_ptr(), _native.gpu_memcpy_d2h)
run("H2D<-pinned ", dev.data_ptr(), pin.data_ptr(), _native.gpu_memcpy_h2d)
run("H2D<-regmmap", dev.data_ptr(), ptr, _native.gpu_memcpy_h2d)
