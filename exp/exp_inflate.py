"""Device-inflate probe: kernel error codes, per-phase timing, host compare."""
import os, sys, time, zlib
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import spark_tfrecord_amd as stf
from spark_tfrecord_amd import _native
from spark_tfrecord_amd.io import paths as P
from spark_tfrecord_amd.engine import gpu as g

os.makedirs("/dev/shm/inf", exist_ok=True)

def probe(tag, data):
    p = f"/dev/shm/inf/{tag}.gz"
    t0 = time.perf_counter()
    gz = P.compress_bytes(data, "gzip")
    t_comp = time.perf_counter() - t0
    with open(p, "wb") as f:
        f.write(gz)
    meta = P.parse_gz_segments_file(p)
    assert meta is not None
    nseg = len(meta[1])
    total_u = sum(u for _, u in meta[1])
    out = torch.empty(max(total_u, 1), dtype=torch.uint8, device="cuda")[:total_u]
    # warm + timed device inflate, err printed
    for it in range(3):
        t0 = time.perf_counter()
        ok = g._device_inflate_group(out, [(p, meta, 0)], torch.device("cuda"))
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        if it == 0 and not ok:
            # re-run with raw err readout
            comp = torch.frombuffer(bytearray(gz), dtype=torch.uint8).cuda()
            body_off, segs, _, _ = meta
            io_, il_, oo_, ol_ = [], [], [], []
            so, uo = body_off, 0
            for c, u in segs:
                io_.append(so); il_.append(c); oo_.append(uo); ol_.append(u)
                so += c; uo += u
            md = torch.as_tensor(np.array([io_, il_, oo_, ol_], np.int64)).cuda()
            err = torch.full((1,), -1, dtype=torch.int64, device="cuda")
            _native.gpu_inflate_segments(comp.data_ptr(), md[0].data_ptr(),
                                         md[1].data_ptr(), md[2].data_ptr(),
                                         md[3].data_ptr(), nseg,
                                         out.data_ptr(), err.data_ptr(),
                                         torch.cuda.current_stream().cuda_stream)
            e = int(err.item())
            print(f"  RAW ERR: seg={(e >> 8) - 0 if e != -1 else -1} "
                  f"cause={e & 0xFF if e != -1 else 0} val={e:#x}")
    t0 = time.perf_counter()
    host = P.decompress_file(p)
    t_host = time.perf_counter() - t0
    match = ok and bytes(out.cpu().numpy().tobytes()) == data
    print(f"{tag:14s} raw={len(data)/1e6:7.1f}MB comp={len(gz)/1e6:7.1f}MB "
          f"nseg={nseg:4d} dev={dt*1000:7.2f}ms ok={ok} match={match} "
          f"host={t_host*1000:7.2f}ms zlib_comp={t_comp*1000:6.0f}ms")

rng = np.random.default_rng(0)
probe("text64M", bytes(rng.integers(65, 90, 64 << 20).astype(np.uint8)))
probe("rand64M", rng.bytes(64 << 20))
semi = b"".join(rng.bytes(200) * 1 for _ in range(100_000))  # framed-ish
probe("semi20M", semi)

# full config-5-like read through the API
payloads = [rng.bytes(200) for _ in range(200_000)]
import pyarrow as pa
t = pa.table({"byteArray": pa.array(payloads, type=pa.large_binary())})
d = "/dev/shm/inf/ds"
stf.write_tfrecord(t, d, record_type="ByteArray", codec="gzip",
                   mode="overwrite", engine="cpu", num_shards=32)
for eng in ("gpu", "cpu"):
    stf.read_tfrecord(d, record_type="ByteArray", engine=eng)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    df = stf.read_tfrecord(d, record_type="ByteArray", engine=eng)
    torch.cuda.synchronize()
    print(f"api read eng={eng}: {time.perf_counter()-t0:.3f}s rows={df.count()}")
