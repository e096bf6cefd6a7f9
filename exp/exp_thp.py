import ctypes, mmap, os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
print("shmem_enabled:", open("/sys/kernel/mm/transparent_hugepage/shmem_enabled").read().strip())
print("enabled:", open("/sys/kernel/mm/transparent_hugepage/enabled").read().strip())
import torch
from spark_tfrecord_amd import _native
torch.cuda.init()
libc = ctypes.CDLL("libc.so.6", use_errno=True)
N = 256 << 20
for use_madv in (False, True):
    path = f"/dev/shm/thp_{use_madv}.bin"
    fd = os.open(path, os.O_RDWR | os.O_CREAT)
    os.ftruncate(fd, N)
    buf = mmap.mmap(fd, N)
    addr = ctypes.addressof(ctypes.c_char.from_buffer(buf))
    if use_madv:
        r = libc.madvise(ctypes.c_void_p(addr), ctypes.c_size_t(N), 14)  # MADV_HUGEPAGE
        print("madvise rc:", r)
    hip = ctypes.CDLL("libamdhip64.so")
    t = time.perf_counter()
    rc = hip.hipHostRegister(ctypes.c_void_p(addr), ctypes.c_size_t(N), 0)
    dt = time.perf_counter() - t
    print(f"madv={use_madv}: hipHostRegister rc={rc} {dt*1000:.1f} ms")
    hip.hipHostUnregister(ctypes.c_void_p(addr))
    del buf
    os.close(fd); os.unlink(path)
