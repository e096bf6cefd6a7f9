"""Measure hipHostRegister cost on fresh tmpfs mmaps + fresh-file write strategies."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from spark_tfrecord_amd import _native
from spark_tfrecord_amd.engine import gpu as g

torch.cuda.init()
dev = torch.zeros(256 << 20, dtype=torch.uint8, device="cuda")
torch.cuda.synchronize()

for mb in (16, 64, 256):
    n = mb << 20
    path = f"/dev/shm/regprobe_{mb}.bin"
    t = time.perf_counter()
    ptr, pinned = _native.file_mmap_pinned(path, n, True)
    t1 = time.perf_counter() - t
    _native.file_mmap_drop(path); os.unlink(path)
    print(f"register fresh {mb}MB: {t1*1000:.1f} ms (pinned={pinned})")

# fresh-file write: mmap-DMA vs staged pwrite, 21MB
img = dev[:21 << 20]
for mode in ("mmap", "staged"):
    ts = []
    for i in range(5):
        path = f"/dev/shm/wprobe_{mode}_{i}.bin"
        t = time.perf_counter()
        if mode == "mmap":
            g.device_to_file(img, path)
        else:
            g._write_file_staged(img, path)
        ts.append(time.perf_counter() - t)
        os.unlink(path)
    print(f"fresh 21MB write {mode}: {min(ts)*1000:.1f} ms")
