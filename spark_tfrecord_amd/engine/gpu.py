"""GPU engine: gfx950 kernel pipeline orchestration.

Decode: sliced H2D (file-mapping DMA) overlapped with the two-pass frame
scan -> chain validation -> fused structure-scan+CRC -> rocprim strided
prefix sums -> extract_fields -> device wire-form columns.
Encode: device wire-form -> size_records -> prefix sum -> record-range
sliced fused emit+CRC overlapped with the D2H DMA into the file mapping.

All kernels launch on the torch current stream, forming one in-order
pipeline; file DMA runs on two side streams gated by events (SURVEY.md §7
step 3). See docs/KERNELS.md for per-kernel design notes and measurements.
"""

from __future__ import annotations

import os
from typing import List, Optional

import numpy as np
import torch

from .. import _native
from ..columnar import RecordBatch, WireColumn, schema_blob, wire_fields
from ..schema import KIND_BYTES, KIND_FLOAT, KIND_INT64, StructType, is_sequence_field, wire_kind_of

FMT = {"Example": _native.FMT_EXAMPLE, "SequenceExample": _native.FMT_SEQUENCE}

_U64_MAX = 0xFFFFFFFFFFFFFFFF


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def _dev(x, dtype, device="cuda") -> torch.Tensor:
    if isinstance(x, torch.Tensor):
        return x.to(device=device, dtype=dtype).contiguous()
    arr = np.ascontiguousarray(x)
    # arrow buffers are read-only and torch warns on wrapping them; the
    # wrapped tensor is only ever read (the .to() below copies to device)
    import warnings

    with warnings.catch_warnings():
        warnings.simplefilter("ignore", UserWarning)
        return torch.as_tensor(arr, dtype=dtype).to(device)


def check_native():
    if not getattr(_native, "HAS_GPU_KERNELS", False):
        raise RuntimeError("_native was built without HIP kernels; rebuild "
                           "with build_native.py before using the GPU engine")


# ---------------------------------------------------------------------------
# Pinned staging buffers: host<->device copies through pageable memory run at
# ~8 GB/s on this platform vs ~57 GB/s pinned (measured, exp/exp_probe.py).
# One reusable pinned buffer per purpose, grown geometrically.
# ---------------------------------------------------------------------------

import threading as _threading

_pinned_tls = _threading.local()


def pinned_buffer(tag: str, nbytes: int) -> torch.Tensor:
    """Reusable pinned staging buffer for `tag`, grown geometrically.

    Pools are THREAD-LOCAL: a tag names a pipeline slot (e.g. the writer's
    rotating D2H buffers), and two user threads writing concurrently must
    not stage through the same memory — one thread's host-side read of the
    view races the other's next async copy into it. Thread-local storage
    also frees a thread's buffers when it exits."""
    pool = getattr(_pinned_tls, "pool", None)
    if pool is None:
        pool = _pinned_tls.pool = {}
    buf = pool.get(tag)
    if buf is None or buf.numel() < nbytes:
        cap = max(nbytes, int((buf.numel() if buf is not None else 1 << 20) * 1.5))
        buf = torch.empty(cap, dtype=torch.uint8, pin_memory=True)
        pool[tag] = buf
    return buf


_CHUNK = 64 << 20  # 64 MiB: overlap file IO with PCIe copies chunkwise
# engine knobs: overridable via env for tuning
_READ_SLICE = int(os.environ.get("TFREC_READ_SLICE", 48 << 20))
_WRITE_SLICES = int(os.environ.get("TFREC_WRITE_SLICES", 8))
# Prescan (structure-scan the arrived prefix under the tail DMA) measured
# NET-NEGATIVE on this host: every host<->device copy executes as a blit
# KERNEL (no SDMA for host-registered or torch-pinned memory — see
# exp/exp_sdma.py), so "overlapped" compute contends with the copy kernels
# and the extra syncs cost more than the hidden scan. Off by default.
_PRESCAN = os.environ.get("TFREC_PRESCAN", "0") == "1"
# msync mapped-DMA writes before the publishing rename (durability parity
# with write_file_atomic's fsync; near-free on tmpfs). TFREC_SYNC=0 disables.
_SYNC_WRITES = os.environ.get("TFREC_SYNC", "1") != "0"


def read_file_to_device(path: str, device="cuda") -> torch.Tensor:
    """File -> HBM. Fast path: hipHostRegister'd mmap of the file, one DMA
    straight out of the page cache (zero host memcpy). Fallback: pinned
    staging chunks through the native pread worker pool."""
    import os as _os

    n = _os.path.getsize(path)
    dev_t = torch.empty(max(n, 1), dtype=torch.uint8, device=device)[:n]
    if n:
        ptr, pinned = _native.file_mmap_pinned(path, n, False)
        if pinned:
            _multi_dma(dev_t.data_ptr(), ptr, n, _native.gpu_memcpy_h2d,
                       after_main=True)
            return dev_t
    return _read_file_staged(path, dev_t)


_copy_streams: Optional[list] = None


def _dma_streams() -> list:
    global _copy_streams
    if _copy_streams is None:
        _copy_streams = [torch.cuda.Stream() for _ in range(2)]
    return _copy_streams


def _multi_dma(dst: int, src: int, n: int, fn, after_main: bool):
    """Issue one big pinned<->HBM copy on up to TWO side streams: measured
    (exp/exp_dma.py) 56 GB/s each way at k<=2 — k=4 overcommits the SDMA
    engines and falls back to ~37 GB/s shader blits. Side streams keep the
    copy off the compute stream. after_main=True orders the copies after
    current-stream work and makes the main stream wait for completion;
    False blocks until the copies land."""
    streams = _dma_streams()
    main = torch.cuda.current_stream()
    k = min(len(streams), max(1, n // (16 << 20)))
    span = (n + k - 1) // k
    for i in range(k):
        o = i * span
        m = min(span, n - o)
        s = streams[i]
        s.wait_stream(main)
        fn(dst + o, src + o, m, s.cuda_stream)
    if after_main:
        for i in range(k):
            main.wait_stream(streams[i])
    else:
        for i in range(k):
            streams[i].synchronize()


def _read_file_staged(path: str, dev: torch.Tensor) -> torch.Tensor:
    import os as _os

    n = dev.numel()
    buf = pinned_buffer("fread", min(n, 2 * _CHUNK) or 1)
    ev = [torch.cuda.Event(), torch.cuda.Event()]
    fd = _os.open(path, _os.O_RDONLY)
    try:
        pos = 0
        which = 0
        while pos < n:
            m = min(_CHUNK, n - pos)
            ev[which].synchronize()  # prior H2D from this half must be done
            _native.pread_parallel(fd, buf.data_ptr() + which * _CHUNK, m, pos)
            dev[pos:pos + m].copy_(buf[which * _CHUNK:which * _CHUNK + m],
                                   non_blocking=True)
            ev[which].record()
            pos += m
            which ^= 1 if n > _CHUNK else 0
    finally:
        _os.close(fd)
    return dev


def device_to_file(img: torch.Tensor, path: str):
    """HBM file image -> file. Fast path: the file is sized, mmap'd and
    hipHostRegister'd, then ONE D2H DMA writes HBM straight into the file's
    page-cache pages (tmpfs: that IS the storage; disk FS: the kernel
    writes back). Fallback: pinned staging + parallel pwrite (a plain
    multi-threaded pwrite to one file serializes on the inode mutex)."""
    import os as _os

    n = img.numel()
    if n == 0:
        open(path, "wb").close()
        return
    # Always map+register: on a FRESH file every write strategy is
    # page-allocation-bound anyway (~6-7 GB/s measured across register+DMA,
    # pwrite pool, mmap memcpy and plain write — exp/exp_freshwrite.py),
    # and the registered mapping makes every REWRITE a pure ~56 GB/s DMA.
    ptr, pinned = _native.file_mmap_pinned(path, n, True)
    if pinned:
        _multi_dma(ptr, img.data_ptr(), n, _native.gpu_memcpy_d2h,
                   after_main=False)
        if _SYNC_WRITES:
            _native.file_mmap_sync(path)
        return
    _write_file_staged(img, path)


def _write_file_staged(img: torch.Tensor, path: str):
    """Fallback writer when hipHostRegister is unavailable: D2H chunks into
    pinned staging overlapped with plain write()s (single-threaded write
    beats a pwrite pool on the per-inode mutex)."""
    import os as _os

    n = img.numel()
    buf = pinned_buffer("fwrite", min(n, 2 * _CHUNK) or 1)
    ev = [torch.cuda.Event(), torch.cuda.Event()]
    fd = _os.open(path, _os.O_RDWR | _os.O_CREAT, 0o644)
    try:
        pos = 0
        which = 0
        # prefetch chunk 0
        m0 = min(_CHUNK, n)
        if n:
            buf[0:m0].copy_(img[0:m0], non_blocking=True)
            ev[0].record()
        while pos < n:
            m = min(_CHUNK, n - pos)
            nxt = pos + m
            nwhich = which ^ (1 if n > _CHUNK else 0)
            if nxt < n:  # start next D2H before blocking on this chunk
                m2 = min(_CHUNK, n - nxt)
                buf[nwhich * _CHUNK:nwhich * _CHUNK + m2].copy_(
                    img[nxt:nxt + m2], non_blocking=True)
                ev[nwhich].record()
            ev[which].synchronize()
            _native.pwrite_parallel(fd, buf.data_ptr() + which * _CHUNK, m, pos)
            pos = nxt
            which = nwhich
        _os.ftruncate(fd, n)
        if _SYNC_WRITES:
            _os.fsync(fd)
    finally:
        _os.close(fd)


# ---------------------------------------------------------------------------
# Prefix sums: rocprim decoupled-lookback scans (native). torch.cumsum's
# innermost-dim kernel launches one block per row — 2.5 ms on a [4, 1M] int64
# scan (profiles/r01_kernel_stats.txt) vs memory-speed here.
# ---------------------------------------------------------------------------

_scan_ws = {}


def _scan_workspace(n: int, device) -> torch.Tensor:
    key = (device.index if hasattr(device, "index") else 0)
    need = _native.gpu_scan_temp_bytes(max(n, 1))
    ws = _scan_ws.get(key)
    if ws is None or ws.numel() < need:
        ws = torch.empty(int(need * 2), dtype=torch.uint8, device=device)
        _scan_ws[key] = ws
    return ws


def _excl_sum_into(out_ptr: int, in_ptr: int, stride: int, n: int, device):
    ws = _scan_workspace(n, device)
    _native.gpu_excl_sum_strided(ws.data_ptr(), ws.numel(), in_ptr, stride,
                                 out_ptr, n, _stream())


def excl_sum(t: torch.Tensor) -> torch.Tensor:
    """Contiguous 1-D int64 tensor -> (n+1)-long exclusive scan, total at [-1]."""
    n = t.numel()
    out = torch.empty(n + 1, dtype=torch.int64, device=t.device)
    _excl_sum_into(out.data_ptr(), t.data_ptr(), 1, n, t.device)
    return out


def device_to_bytes(img: torch.Tensor) -> bytes:
    n = img.numel()
    buf = pinned_buffer("d2h", n)
    buf[:n].copy_(img, non_blocking=True)
    torch.cuda.synchronize()
    return buf.numpy()[:n].tobytes()


def device_to_pinned_view(img: torch.Tensor, tag: str = "d2h_img") -> np.ndarray:
    """One link-speed D2H of a device image into the reusable pinned buffer;
    returns a numpy VIEW of it (valid until the tag's next use). Partitioned
    writes slice this view per partition file — registering a fresh mmap per
    part file costs ~0.16 ms/MB in hipHostRegister alone, ~10x the bytes'
    DMA time for many small fresh files (r01 config-3 cliff)."""
    n = img.numel()
    buf = pinned_buffer(tag, n)
    buf[:n].copy_(img, non_blocking=True)
    torch.cuda.synchronize()
    return buf.numpy()[:n]


# ---------------------------------------------------------------------------
# GPU frame scan: parallel frame-boundary discovery + chain validation.
# ---------------------------------------------------------------------------

def scan_frames_device(data: torch.Tensor):
    """Device file image -> (payload_off, payload_len) device tensors.

    Every byte position is CRC-tested as a candidate frame head in parallel
    (two-pass count/emit — candidates come out position-sorted by
    construction, no device sort); the chain pos[k+1] == pos[k]+16+len[k]
    must hold exactly — a false positive (p=2^-32/byte) or a torn file
    breaks it and raises."""
    check_native()
    N = data.numel()
    device = data.device
    if N == 0:
        z = torch.zeros(0, dtype=torch.int64, device=device)
        return z, z.clone()
    B = _native.gpu_frame_scan_blocks(0, N)
    block_counts = torch.empty(B, dtype=torch.int64, device=device)
    _native.gpu_frame_scan_count(data.data_ptr(), N, 0, N, 0,
                                 block_counts.data_ptr(), _stream())
    off, lens, _ = _emit_and_chain(data, [(0, N)], block_counts, N)
    return off, lens


def _prescan_prefix(data, pre_ranges, block_counts, nblocks, N, arrived,
                    schema, record_type):
    """Emit the candidates of the ALREADY-ARRIVED slice prefix and launch
    the structure scan for every record that lies entirely within `arrived`
    bytes — on the main stream, which at this point is NOT yet ordered
    after the in-flight tail slices, so this work overlaps their DMA.
    Returns (stats1[K,F,6], err1, K) for decode_device, or None."""
    device = data.device
    nb_pre = sum(nblocks[:len(pre_ranges)])
    if nb_pre == 0:
        return None
    block_off = excl_sum(block_counts[:nb_pre])
    C = int(block_off[-1].item())  # syncs main: prefix counts only
    if C == 0:
        return None
    pos = torch.empty(C, dtype=torch.int64, device=device)
    lens = torch.empty(C, dtype=torch.int64, device=device)
    base = 0
    for s, e in pre_ranges:
        _native.gpu_frame_scan_emit(data.data_ptr(), N, s, e, base,
                                    block_off.data_ptr(), pos.data_ptr(),
                                    lens.data_ptr(), _stream())
        base += _native.gpu_frame_scan_blocks(s, e)
    # records fully inside the arrived prefix (candidates are sorted)
    K = int(torch.searchsorted(pos + 16 + lens, torch.tensor(
        arrived + 1, device=device)).item())
    if K == 0:
        return None
    fields = wire_fields(schema)
    F = len(fields)
    if F == 0:
        return None
    blob = torch.frombuffer(bytearray(schema_blob(schema)),
                            dtype=torch.uint8).to(device)
    stats1 = torch.empty((K, F, 6), dtype=torch.int64, device=device)
    err1 = torch.zeros(2, dtype=torch.int32, device=device)
    off1 = (pos[:K] + 12).contiguous()
    lens1 = lens[:K].contiguous()
    _native.gpu_scan_records(data.data_ptr(), off1.data_ptr(),
                             lens1.data_ptr(), K, FMT[record_type],
                             blob.data_ptr(), F, stats1.data_ptr(),
                             err1.data_ptr(), 0, _stream())
    return stats1, err1, K


def _emit_and_chain(data, ranges, block_counts, N):
    """Prefix-sum the per-block candidate counts, emit the (sorted)
    candidates at exact offsets, then validate the frame chain. Returns
    (payload_off, payload_len, clean) — clean=False when a false-positive
    candidate had to be stitched out on host."""
    device = data.device
    block_off = excl_sum(block_counts)
    C = int(block_off[-1].item())
    if C == 0:
        raise RuntimeError("corrupt TFRecord: no valid frame header found")
    pos = torch.empty(C, dtype=torch.int64, device=device)
    lens = torch.empty(C, dtype=torch.int64, device=device)
    base = 0
    for s, e in ranges:
        _native.gpu_frame_scan_emit(data.data_ptr(), N, s, e, base,
                                    block_off.data_ptr(), pos.data_ptr(),
                                    lens.data_ptr(), _stream())
        base += _native.gpu_frame_scan_blocks(s, e)
    # chain check fused to ONE device scalar -> one sync
    expect_next = pos + 16 + lens
    ok = bool(((pos[0] == 0) & (expect_next[-1] == N)
               & (expect_next[:-1] == pos[1:]).all()).item())
    if not ok:
        # rare: false-positive candidate inside a payload — stitch on host
        pos_h = pos.cpu().numpy()
        len_h = lens.cpu().numpy()
        import numpy as _np

        keep = []
        cur = 0
        idx = 0
        pmap = {int(p): i for i, p in enumerate(pos_h)}
        while cur < N:
            i = pmap.get(cur)
            if i is None:
                raise RuntimeError(
                    f"corrupt TFRecord: broken frame chain at offset {cur}")
            keep.append(i)
            cur = int(pos_h[i] + 16 + len_h[i])
        sel = torch.as_tensor(_np.asarray(keep, _np.int64), device=device)
        pos = pos[sel]
        lens = lens[sel]
    return pos + 12, lens, ok


# ---------------------------------------------------------------------------
# Decode
# ---------------------------------------------------------------------------

def crc_verify_device(data: torch.Tensor, off: torch.Tensor, lens: torch.Tensor,
                      avg_bytes: int = 0):
    """Raise on any bad frame CRC (parallel over records; records larger
    than ~8 KB verify one-per-wavefront with GF(2)-combined chunk CRCs)."""
    R = off.numel()
    if R == 0:
        return
    if not avg_bytes:
        avg_bytes = max(0, (data.numel() // R) - 16)
    err = torch.full((1,), -1, dtype=torch.int64, device=data.device)  # all-ones
    _native.gpu_crc_verify(data.data_ptr(), off.data_ptr(), lens.data_ptr(), R,
                           err.data_ptr(), _stream(), avg_bytes)
    bad = int(err.item())  # syncs
    if bad != -1:
        raise RuntimeError(
            f"corrupt TFRecord: bad CRC in record {bad - 1 if bad > 0 else bad}")


def decode_device(data: torch.Tensor, off: torch.Tensor, lens: torch.Tensor,
                  schema: StructType, record_type: str,
                  verify_crc: bool = True, prescan=None) -> RecordBatch:
    """Decode framed records already resident in HBM into device wire-form.
    `prescan` = (stats1, err1, K) from _prescan_prefix: rows [0, K) were
    already structure-scanned while the tail of the file was in DMA flight."""
    check_native()
    device = data.device
    R = off.numel()
    fields = wire_fields(schema)
    F = len(fields)
    if verify_crc and record_type == "ByteArray":
        crc_verify_device(data, off, lens)

    if record_type == "ByteArray":
        dst_off = excl_sum(lens.contiguous())
        total = int(dst_off[-1].item())
        out = torch.empty(total, dtype=torch.uint8, device=device)
        if R:
            _native.gpu_gather_payloads(data.data_ptr(), off.data_ptr(),
                                        lens.data_ptr(), dst_off.data_ptr(), R,
                                        out.data_ptr(), _stream(), total // R)
        col = WireColumn(kind=KIND_BYTES, is_seq=False,
                         presence=torch.ones(R, dtype=torch.uint8, device=device),
                         row_off=torch.arange(R + 1, dtype=torch.int64, device=device),
                         values=out, elem_off=dst_off)
        return RecordBatch(schema, [col], R)

    if F == 0 or R == 0:
        cols = [WireColumn(wire_kind_of(f.dataType), is_sequence_field(f.dataType),
                           torch.zeros(R, dtype=torch.uint8, device=device),
                           torch.zeros(R + 1, dtype=torch.int64, device=device),
                           torch.zeros(0, dtype=torch.int64, device=device))
                for f in fields]
        return RecordBatch(schema, cols, R)

    blob = torch.frombuffer(bytearray(schema_blob(schema)),
                            dtype=torch.uint8).to(device)
    # FieldStat[R][F] as int64 [R,F,6]: pos,len,nvals,nbytes,nlists,(kind|err)
    stats = torch.empty((R, F, 6), dtype=torch.int64, device=device)
    # err buffers are int32[2]: (codec error code, 1 + offending record index)
    err = torch.zeros(2, dtype=torch.int32, device=device)
    # frame CRC verification is FUSED into the structure scan: the record
    # bytes are CRC'd while L2-hot from the parse. (A concurrent side-stream
    # CRC pass was tried and reverted: every host<->device copy on this
    # stack is a blit KERNEL, so "free" overlap doesn't exist and the fused
    # form has ~1 ms less total GPU work per 215 MB.)
    # NOTE: for Example/SequenceExample the CRC stays FUSED into the scan at
    # any record size — the serial protobuf parse dominates huge records, so
    # a separate wave-CRC pass only adds a second full read (measured).
    # The wave CRC/copy kernels serve the parse-free ByteArray paths.
    crc_err = None
    fuse_crc = verify_crc
    if fuse_crc:
        crc_err = torch.full((1,), -1, dtype=torch.int64, device=device)
    r0 = 0
    err1 = None
    if prescan is not None:
        stats1, err1, K = prescan
        if K <= R and stats1.shape[1] == F:
            stats[:K].copy_(stats1)
            r0 = K
        else:
            err1 = None
    # average record size steers the one-per-lane vs one-per-wave kernels
    avg_bytes = max(0, (data.numel() // R) - 16) if R else 0
    if R > r0:
        stride = F * 6 * 8
        _native.gpu_scan_records(data.data_ptr(), off.data_ptr() + r0 * 8,
                                 lens.data_ptr() + r0 * 8, R - r0,
                                 FMT[record_type], blob.data_ptr(), F,
                                 stats.data_ptr() + r0 * stride,
                                 err.data_ptr(),
                                 crc_err.data_ptr() if fuse_crc else 0,
                                 _stream(), avg_bytes)
    crc_err_pre = None
    if fuse_crc and r0 > 0:
        # prescanned rows skipped the fused path: verify them separately.
        # Separate error buffer — the fused kernel reports indices relative
        # to its r0-based sub-launch, this one reports absolute indices.
        crc_err_pre = torch.full((1,), -1, dtype=torch.int64, device=device)
        _native.gpu_crc_verify(data.data_ptr(), off.data_ptr(),
                               lens.data_ptr(), r0, crc_err_pre.data_ptr(),
                               _stream(), 0)

    # Per-field exclusive prefix sums ([F, R+1] x 3 planes) in ONE rocprim
    # launch: a head-flag segmented scan over the [F*R] triple sequence
    # scatters into all planes (3F separate scans cost 6F launches).
    val_base = torch.empty((F, R + 1), dtype=torch.int64, device=device)
    byte_base = torch.empty((F, R + 1), dtype=torch.int64, device=device)
    list_base = torch.empty((F, R + 1), dtype=torch.int64, device=device)
    val_base[:, 0] = 0
    byte_base[:, 0] = 0
    list_base[:, 0] = 0
    key = ("stat", device.index if hasattr(device, "index") else 0)
    need = _native.gpu_stat_scan_temp_bytes(R, F)
    ws = _scan_ws.get(key)
    if ws is None or ws.numel() < need:
        ws = torch.empty(int(need * 2) or 1, dtype=torch.uint8, device=device)
        _scan_ws[key] = ws
    _native.gpu_stat_scans(ws.data_ptr(), ws.numel(), stats.data_ptr(), R, F,
                           val_base.data_ptr(), byte_base.data_ptr(),
                           list_base.data_ptr(), _stream())
    totals = torch.stack([val_base[:, -1], byte_base[:, -1], list_base[:, -1]])
    totals_h = totals.cpu()  # one sync for all allocations
    if fuse_crc:
        bad = int(crc_err.item())
        # fused-scan indices are relative to the r0-based sub-launch
        bad_abs = (bad - 1 + r0) if bad != -1 else -1
        if crc_err_pre is not None:
            bp = int(crc_err_pre.item())
            if bp != -1:
                bad_abs = bp - 1 if bad_abs == -1 else min(bad_abs, bp - 1)
        if bad_abs != -1:
            raise RuntimeError(
                f"corrupt TFRecord: bad CRC in record {bad_abs}")
    rc = int(err[0].item())
    rec = int(err[1].item()) - 1 + r0  # scan indices are relative to r0
    if rc == 0 and err1 is not None:
        rc = int(err1[0].item())
        rec = int(err1[1].item()) - 1  # prescan launch base is record 0
    if rc != 0:
        where = f" in record {rec}" if rec >= 0 else ""
        raise RuntimeError(f"TFRecord decode failed{where} (native error "
                           f"{rc}; kind mismatch or malformed record)")

    metas = []
    outs = []
    for i, f in enumerate(fields):
        kind = wire_kind_of(f.dataType)
        seq = is_sequence_field(f.dataType)
        nv = int(totals_h[0, i])
        nb = int(totals_h[1, i])
        nl = int(totals_h[2, i])
        o = {"kind": kind, "seq": seq}
        o["i64_vals"] = torch.empty(nv if kind == KIND_INT64 else 0,
                                    dtype=torch.int64, device=device)
        o["f32_vals"] = torch.empty(nv if kind == KIND_FLOAT else 0,
                                    dtype=torch.float32, device=device)
        o["bytes_data"] = torch.empty(nb, dtype=torch.uint8, device=device)
        o["elem_len"] = torch.empty(nv if kind == KIND_BYTES else 0,
                                    dtype=torch.int64, device=device)
        o["sub_count"] = torch.empty(nl, dtype=torch.int64, device=device)
        outs.append(o)
        metas.append({
            "kind": kind, "is_seq": 1 if seq else 0,
            "i64_vals": o["i64_vals"].data_ptr(),
            "f32_vals": o["f32_vals"].data_ptr(),
            "bytes_data": o["bytes_data"].data_ptr(),
            "elem_len": o["elem_len"].data_ptr(),
            "sub_count": o["sub_count"].data_ptr(),
            "val_base": val_base[i].data_ptr(),
            "byte_base": byte_base[i].data_ptr(),
            "list_base": list_base[i].data_ptr() if seq else 0,
        })

    meta_dev = torch.empty(F * _native.gpu_devmeta_bytes(), dtype=torch.uint8,
                           device=device)
    _native.gpu_extract_fields(data.data_ptr(), R, F, stats.data_ptr(), metas,
                               meta_dev.data_ptr(), err.data_ptr(), _stream(),
                               avg_bytes)

    cols: List[WireColumn] = []
    for i, (f, o) in enumerate(zip(fields, outs)):
        kind = o["kind"]
        seq = o["seq"]
        presence = (stats[:, i, 0] >= 0).to(torch.uint8)
        elem_off = None
        if kind == KIND_BYTES:
            elem_off = excl_sum(o["elem_len"])
        sub_off = None
        if seq:
            sub_off = excl_sum(o["sub_count"])
        values = (o["i64_vals"] if kind == KIND_INT64 else
                  o["f32_vals"] if kind == KIND_FLOAT else o["bytes_data"])
        cols.append(WireColumn(kind, seq, presence, val_base[i], values,
                               elem_off, list_base[i] if seq else None, sub_off))
    if int(err[0].item()) != 0:
        rec = int(err[1].item()) - 1
        where = f" (record {rec})" if rec >= 0 else ""
        raise RuntimeError(f"TFRecord decode failed in value extraction"
                           f"{where} (native error {int(err[0].item())})")
    return RecordBatch(schema, cols, R)


def decode_buffer_device(data_np: np.ndarray, schema: StructType, record_type: str,
                         verify_crc: bool = True, device="cuda") -> RecordBatch:
    """Host bytes -> device batch (frame discovery AND decode on the GPU)."""
    data_np = np.ascontiguousarray(data_np, np.uint8)
    n = data_np.size
    stage = pinned_buffer("h2d", n)
    stage.numpy()[:n] = data_np
    data = torch.empty(n, dtype=torch.uint8, device=device)
    data.copy_(stage[:n], non_blocking=True)
    off, lens = scan_frames_device(data)
    return decode_device(data, off, lens, schema, record_type, verify_crc)


def read_file_to_batch(path: str, schema: StructType, record_type: str,
                       verify_crc: bool = True, device="cuda") -> RecordBatch:
    """Uncompressed file -> device batch (H2D sliced + overlapped with the
    frame scan; see read_file_to_batch_pipelined)."""
    return read_file_to_batch_pipelined(path, schema, record_type, verify_crc,
                                        device)


# ---------------------------------------------------------------------------
# Encode
# ---------------------------------------------------------------------------

def _col_ptrs(col: WireColumn) -> dict:
    def p(t):
        return t.data_ptr() if isinstance(t, torch.Tensor) and t.numel() else 0

    return {
        "kind": col.kind, "is_seq": 1 if col.is_seq else 0,
        "presence": p(col.presence), "row_off": p(col.row_off),
        "list_off": p(col.list_off) if col.list_off is not None else 0,
        "sub_off": p(col.sub_off) if col.sub_off is not None else 0,
        "elem_off": p(col.elem_off) if col.elem_off is not None else 0,
        "values_bytes": p(col.values) if col.kind == KIND_BYTES else 0,
        "values_i64": p(col.values) if col.kind == KIND_INT64 else 0,
        "values_f32": p(col.values) if col.kind == KIND_FLOAT else 0,
    }


def _check_emit_err(err: torch.Tensor):
    if int(err[0].item()) != 0:
        rec = int(err[1].item()) - 1
        where = f" in record {rec}" if rec >= 0 else ""
        raise RuntimeError(f"TFRecord encode failed: size/emit mismatch{where}")


def encode_device(batch: RecordBatch, record_type: str) -> torch.Tensor:
    """Device wire-form batch -> framed file image as a device u8 tensor."""
    check_native()
    R = batch.num_rows
    device = batch.columns[0].presence.device if batch.columns else torch.device("cuda")
    if record_type == "ByteArray":
        col = batch.columns[0]
        lens = col.elem_off[1:] - col.elem_off[:-1]
        frame_off = excl_sum((lens + 16).contiguous())
        total = int(frame_off[-1].item())
        file = torch.empty(total, dtype=torch.uint8, device=device)
        if R:
            _native.gpu_frame_bytes(col.values.data_ptr(), col.elem_off.data_ptr(),
                                    frame_off.data_ptr(), R, file.data_ptr(),
                                    _stream(), (total // R) - 16)
        return file

    blob = torch.frombuffer(bytearray(schema_blob(batch.schema)),
                            dtype=torch.uint8).to(device)
    col_dicts = [_col_ptrs(c) for c in batch.columns]
    cols_dev = torch.empty(_native.gpu_devcols_bytes(len(col_dicts)),
                           dtype=torch.uint8, device=device)
    psize = torch.empty(R, dtype=torch.int64, device=device)
    _native.gpu_size_records(col_dicts, cols_dev.data_ptr(), blob.data_ptr(),
                             FMT[record_type], R, psize.data_ptr(), _stream())
    frame_off = excl_sum(psize)
    total = int(frame_off[-1].item())
    file = torch.empty(total, dtype=torch.uint8, device=device)
    err = torch.zeros(2, dtype=torch.int32, device=device)
    _native.gpu_emit_records(cols_dev.data_ptr(), blob.data_ptr(),
                             FMT[record_type], 0, R, frame_off.data_ptr(),
                             file.data_ptr(), err.data_ptr(), _stream(),
                             total // max(R, 1))
    _check_emit_err(err)
    return file


def write_batch_to_file(batch: RecordBatch, path: str,
                        record_type: str = "Example",
                        slices: Optional[int] = None) -> int:
    """Device batch -> framed TFRecord file, with the emit kernel sliced over
    record ranges so the D2H DMA of slice k streams to the file's mapped
    pages while slice k+1 is still being emitted (SURVEY.md §7 step 3:
    double-buffered transport overlapped with encode). Returns file bytes."""
    check_native()
    R = batch.num_rows
    if record_type == "ByteArray" or R == 0:
        img = encode_device(batch, record_type)
        device_to_file(img, path)
        return img.numel()
    device = batch.columns[0].presence.device
    blob = torch.frombuffer(bytearray(schema_blob(batch.schema)),
                            dtype=torch.uint8).to(device)
    col_dicts = [_col_ptrs(c) for c in batch.columns]
    cols_dev = torch.empty(_native.gpu_devcols_bytes(len(col_dicts)),
                           dtype=torch.uint8, device=device)
    psize = torch.empty(R, dtype=torch.int64, device=device)
    _native.gpu_size_records(col_dicts, cols_dev.data_ptr(), blob.data_ptr(),
                             FMT[record_type], R, psize.data_ptr(), _stream())
    frame_off = excl_sum(psize)
    S = max(1, min(slices if slices is not None else _WRITE_SLICES, R))
    ridx = [R * s // S for s in range(S + 1)]
    bounds = frame_off[ridx].cpu()  # one sync: slice byte bounds + total
    total = int(bounds[-1])
    file = torch.empty(total, dtype=torch.uint8, device=device)
    err = torch.zeros(2, dtype=torch.int32, device=device)
    ptr, pinned = _native.file_mmap_pinned(path, total, True)
    if not pinned:
        _native.gpu_emit_records(cols_dev.data_ptr(), blob.data_ptr(),
                                 FMT[record_type], 0, R, frame_off.data_ptr(),
                                 file.data_ptr(), err.data_ptr(), _stream(),
                                 total // R)
        _check_emit_err(err)
        _write_file_staged(file, path)
        return total
    main = torch.cuda.current_stream()
    streams = _dma_streams()
    for s in range(S):
        _native.gpu_emit_records(cols_dev.data_ptr(), blob.data_ptr(),
                                 FMT[record_type], ridx[s], ridx[s + 1],
                                 frame_off.data_ptr(), file.data_ptr(),
                                 err.data_ptr(), _stream(), total // R)
        b0, b1 = int(bounds[s]), int(bounds[s + 1])
        if b1 > b0:
            w = streams[s % len(streams)]
            w.wait_stream(main)
            _native.gpu_memcpy_d2h(ptr + b0, file.data_ptr() + b0, b1 - b0,
                                   w.cuda_stream)
    for w in streams:
        w.synchronize()
    if _SYNC_WRITES:
        _native.file_mmap_sync(path)
    _check_emit_err(err)
    return total


def read_file_to_batch_pipelined(path: str, schema: StructType, record_type: str,
                                 verify_crc: bool = True,
                                 device="cuda") -> RecordBatch:
    """File -> device batch with the H2D DMA sliced so the frame-candidate
    scan of slice k runs while slice k+1 is still in flight. Positions within
    32 bytes of a slice boundary are deferred to the next slice's launch
    (their load window extends past the boundary)."""
    import os as _os

    check_native()
    n = _os.path.getsize(path)
    if n == 0:
        z = torch.zeros(0, dtype=torch.int64, device=device)
        return decode_device(torch.zeros(0, dtype=torch.uint8, device=device),
                             z, z.clone(), schema, record_type, verify_crc)
    ptr, pinned = _native.file_mmap_pinned(path, n, False)
    if not pinned:
        data = _read_file_staged(path, torch.empty(n, dtype=torch.uint8,
                                                   device=device))
        off, lens = scan_frames_device(data)
        return decode_device(data, off, lens, schema, record_type, verify_crc)
    data = torch.empty(n, dtype=torch.uint8, device=device)
    span = _READ_SLICE
    S = max(1, (n + span - 1) // span)
    # pre-size the per-block count buffer over the slice ranges
    ranges = []
    scanned = 0
    for s in range(S):
        b1 = min(n, (s + 1) * span)
        scan_end = b1 - 32 if s < S - 1 else n
        if scan_end > scanned:
            ranges.append((scanned, scan_end))
            scanned = scan_end
    nblocks = [int(_native.gpu_frame_scan_blocks(a, b)) for a, b in ranges]
    block_counts = torch.empty(max(sum(nblocks), 1), dtype=torch.int64,
                               device=device)
    main = torch.cuda.current_stream()
    streams = _dma_streams()
    # issue ALL H2D slices up front on the side streams; per-slice events
    # let the main stream gate each count pass on just its own slice
    events = []
    for st in streams:
        st.wait_stream(main)  # `data` allocation ordering
    for s in range(S):
        b0, b1 = s * span, min(n, (s + 1) * span)
        st = streams[s % len(streams)]
        _native.gpu_memcpy_h2d(data.data_ptr() + b0, ptr + b0, b1 - b0,
                               st.cuda_stream)
        ev = torch.cuda.Event()
        ev.record(st)
        events.append(ev)
    data.record_stream(streams[0])
    if len(streams) > 1:
        data.record_stream(streams[1])

    def launch_counts(ri: int, base: int):
        _native.gpu_frame_scan_count(data.data_ptr(), n, ranges[ri][0],
                                     ranges[ri][1], base,
                                     block_counts.data_ptr(), _stream())

    # stage point: once ~2/3 of the slices have landed, scan the records that
    # are fully inside the arrived prefix WHILE the tail slices are still in
    # DMA flight (hides most of the structure-scan under the H2D)
    k0 = ((2 * S) // 3 if _PRESCAN and S >= 3 and record_type != "ByteArray"
          else 0)

    ri = 0
    base = 0

    def consume_slices(upto: int):
        nonlocal ri, base
        for s in range(upto):
            if events[s] is None:
                continue
            main.wait_event(events[s])
            events[s] = None
            scan_end = min(n, (s + 1) * span) - 32 if s < S - 1 else n
            if ri < len(ranges) and ranges[ri][1] == scan_end:
                launch_counts(ri, base)
                base += nblocks[ri]
                ri += 1

    prescan = None
    if k0:
        consume_slices(k0)  # main now ordered after the first k0 slices ONLY
        prescan = _prescan_prefix(data, ranges[:ri], block_counts, nblocks, n,
                                  arrived=min(n, k0 * span), schema=schema,
                                  record_type=record_type)
    consume_slices(S)
    off, lens, clean = _emit_and_chain(data, ranges, block_counts, n)
    if not clean:
        prescan = None  # host-stitched chain: prescanned rows are stale
    return decode_device(data, off, lens, schema, record_type, verify_crc,
                         prescan=prescan)


def gz_device_meta(path: str):
    """Segment table of OUR gzip files (None for foreign/table-less gzip):
    gates the device-inflate read path."""
    from ..io import paths as P

    if not getattr(_native, "HAS_GPU_KERNELS", False):
        return None
    return P.parse_gz_segments_file(path)


def _device_inflate_group(data: torch.Tensor, gz_items, device) -> bool:
    """Inflate several gzip files' full-flush segments in ONE kernel launch
    (one segment per lane, csrc/hip/inflate.hip), each file's output landing
    at its slice of `data`. gz_items = [(path, parse_gz_segments_file meta,
    out_base)]. Compressed bodies DMA straight from the page cache (pinned
    mmaps). Returns False when the kernel reported any malformed segment —
    the caller redoes those files on host zlib. The gzip CRC32 trailer is
    NOT checked here: the TFRecord layer CRC32C-verifies every record of the
    inflated bytes (a corrupt stream also breaks the frame chain)."""
    import os as _os

    comp_sizes = [_os.path.getsize(p) for p, _, _ in gz_items]
    comp_total = sum(comp_sizes)
    comp = torch.empty(max(comp_total, 1), dtype=torch.uint8,
                       device=device)[:comp_total]
    main = torch.cuda.current_stream()
    streams = _dma_streams()
    in_off = []
    in_len = []
    out_off = []
    out_len = []
    used = []
    coff = 0
    for k, (p, meta, out_base) in enumerate(gz_items):
        body_off, segs, _crc, _isize = meta
        ptr, pinned = _native.file_mmap_pinned(p, comp_sizes[k], False)
        if pinned:
            st = streams[k % len(streams)]
            st.wait_stream(main)
            _native.gpu_memcpy_h2d(comp.data_ptr() + coff, ptr, comp_sizes[k],
                                   st.cuda_stream)
            used.append(st)
        else:
            _read_file_staged(p, comp[coff:coff + comp_sizes[k]])
        so = coff + body_off
        uo = out_base
        for c, u in segs:
            in_off.append(so)
            in_len.append(c)
            out_off.append(uo)
            out_len.append(u)
            so += c
            uo += u
        coff += comp_sizes[k]
    for st in set(used):
        main.wait_stream(st)
    # Route each segment to the kernel shape its data favors (measured,
    # profiles/RESULTS.md): literal-heavy segments — compression ratio near
    # 1, every output byte is one Huffman symbol — run ~20% faster TWO per
    # wave (half-wave streams share the same vector instructions); match-
    # heavy segments favor the whole-wave kernel (wide cooperative copies,
    # no divergence between halves). TFREC_INFLATE_STREAMS=1|2 forces one.
    meta_np = np.array([in_off, in_len, out_off, out_len], np.int64)
    force = _os.environ.get("TFREC_INFLATE_STREAMS", "")
    spw_lit = 2
    if force in ("1", "2", "4"):
        spw_lit = int(force)
        lit = np.full(meta_np.shape[1], spw_lit > 1)
    else:
        lit = meta_np[1] >= (0.85 * np.maximum(meta_np[3], 1))
    err = torch.full((1,), -1, dtype=torch.int64, device=device)
    for mask, spw in ((lit, spw_lit), (~lit, 1)):
        k = int(mask.sum())
        if k == 0:
            continue
        idx = np.nonzero(mask)[0]
        if spw == 2 and k > 2:
            # pair like-sized segments in one wave: halves reconverge at
            # segment end, so a short partner idles under a long one
            idx = idx[np.argsort(-meta_np[3, idx], kind="stable")]
        sub = torch.as_tensor(
            np.ascontiguousarray(meta_np[:, idx])).to(device)
        _native.gpu_inflate_segments(
            comp.data_ptr(), sub[0].data_ptr(), sub[1].data_ptr(),
            sub[2].data_ptr(), sub[3].data_ptr(), k,
            data.data_ptr(), err.data_ptr(), _stream(), spw)
    return int(err.item()) == -1


def read_gzip_file_to_device(path: str, device="cuda") -> Optional[torch.Tensor]:
    """Our-format gzip file -> decompressed bytes in HBM via the device
    inflater. None when the file carries no segment table or the kernel
    rejected a segment (callers fall back to host zlib)."""
    meta = gz_device_meta(path)
    if meta is None:
        return None
    total_u = sum(u for _, u in meta[1])
    data = torch.empty(max(total_u, 1), dtype=torch.uint8,
                       device=device)[:total_u]
    if not _device_inflate_group(data, [(path, meta, 0)], device):
        return None
    return data


def read_files_to_batch(paths, schema: StructType, record_type: str,
                        verify_crc: bool = True, device="cuda"):
    """Decode MANY files as ONE GPU pipeline: their (decompressed) images
    are concatenated in HBM (TFRecord frames are concatenable, so the frame
    chain spans file boundaries exactly), scanned and decoded once, and the
    per-file row counts recovered with a searchsorted over frame offsets.
    Uncompressed files DMA straight from the page cache; gzip files bearing
    our segment table are inflated by the device inflater (all files'
    segments in one launch). Returns (device RecordBatch, np.ndarray
    row_counts per file)."""
    import os as _os

    from ..io import paths as P

    check_native()
    metas = []
    sizes = []
    for p in paths:
        if P.codec_from_path(p) == "gzip":
            meta = P.parse_gz_segments_file(p)
            if meta is None:
                raise ValueError(f"gzip file without segment table: {p} "
                                 "(host path required)")
            metas.append(meta)
            sizes.append(sum(u for _, u in meta[1]))
        else:
            metas.append(None)
            sizes.append(_os.path.getsize(p))
    n = sum(sizes)
    if n == 0:
        if schema is None:
            raise ValueError(
                "Could not infer schema: no non-empty TFRecord files found")
        z = torch.zeros(0, dtype=torch.int64, device=device)
        return (decode_device(torch.zeros(0, dtype=torch.uint8, device=device),
                              z, z.clone(), schema, record_type, verify_crc),
                np.zeros(len(paths), np.int64))
    data = torch.empty(n, dtype=torch.uint8, device=device)
    bounds = np.zeros(len(paths) + 1, np.int64)
    np.cumsum(sizes, out=bounds[1:])
    main = torch.cuda.current_stream()
    streams = _dma_streams()
    gz_items = []
    for i, p in enumerate(paths):
        if not sizes[i]:
            continue
        if metas[i] is not None:
            gz_items.append((p, metas[i], int(bounds[i])))
            continue
        ptr, pinned = _native.file_mmap_pinned(p, sizes[i], False)
        if pinned:
            st = streams[i % len(streams)]
            _native.gpu_memcpy_h2d(data.data_ptr() + int(bounds[i]), ptr,
                                   sizes[i], st.cuda_stream)
            main.wait_stream(st)
        else:
            _read_file_staged(p, data[int(bounds[i]):int(bounds[i + 1])])
    if gz_items and not _device_inflate_group(data, gz_items, device):
        # rare fallback: a segment the kernel rejected — host zlib fills
        # the affected files' slices
        import logging

        logging.getLogger(__name__).warning(
            "device inflate rejected a segment; host zlib fallback for "
            "%d gzip file(s)", len(gz_items))
        for p, meta, base in gz_items:
            blob = np.frombuffer(P.decompress_file(p), np.uint8)
            stage = pinned_buffer("gzfb", blob.size)
            stage.numpy()[:blob.size] = blob
            data[base:base + blob.size].copy_(stage[:blob.size],
                                              non_blocking=True)
            # the staging buffer is reused next iteration: wait out the copy
            torch.cuda.current_stream().synchronize()
    off, lens = scan_frames_device(data)
    frame_start = off - 12
    counts = torch.searchsorted(
        frame_start, torch.as_tensor(bounds, device=device)).cpu().numpy()
    row_counts = np.diff(counts)
    if schema is None:
        # FUSED schema inference: the reference scans the first non-empty
        # file in a separate job (DefaultSource.scala:31-39); here the file
        # image is already in HBM, so the lattice kernel runs over that
        # file's frames and the decode reuses the same image — a
        # schema-less read costs ONE pass instead of two.
        from ..infer import byte_array_schema, schema_from_codes

        if record_type == "ByteArray":
            schema = byte_array_schema()
        else:
            i = next((k for k in range(len(paths)) if row_counts[k] > 0), None)
            if i is None:
                raise ValueError(
                    "Could not infer schema: no non-empty TFRecord files found")
            r0, r1 = int(counts[i]), int(counts[i + 1])
            codes = infer_codes_device(data, off[r0:r1].contiguous(),
                                       lens[r0:r1].contiguous(), record_type)
            schema = schema_from_codes(codes)
    batch = decode_device(data, off, lens, schema, record_type, verify_crc)
    return batch, row_counts


def encode_partitions_device(batch: RecordBatch, part_codes: np.ndarray,
                             num_parts: int, record_type: str):
    """Encode ALL rows once, then split into per-partition file images by
    GATHERING framed records (TFRecord frames are concatenable, so a
    partitioned write never re-serializes a row — the partition step is a
    pure byte gather in HBM). Returns (image, [(part_code, lo, hi), ...])
    where image[lo:hi] is partition part_code's complete file image.

    The reference delegates this grouping to Spark's FileFormatWriter sort
    (SURVEY.md §3.3); here the grouping is a device argsort of partition
    codes + one gather kernel pass."""
    check_native()
    R = batch.num_rows
    device = batch.columns[0].presence.device if batch.columns else torch.device("cuda")
    blob = torch.frombuffer(bytearray(schema_blob(batch.schema)),
                            dtype=torch.uint8).to(device)
    col_dicts = [_col_ptrs(c) for c in batch.columns]
    cols_dev = torch.empty(_native.gpu_devcols_bytes(len(col_dicts)),
                           dtype=torch.uint8, device=device)
    psize = torch.empty(R, dtype=torch.int64, device=device)
    _native.gpu_size_records(col_dicts, cols_dev.data_ptr(), blob.data_ptr(),
                             FMT[record_type], R, psize.data_ptr(), _stream())
    frame_off = excl_sum(psize)
    total = int(frame_off[-1].item())
    file = torch.empty(total, dtype=torch.uint8, device=device)
    err = torch.zeros(2, dtype=torch.int32, device=device)
    _native.gpu_emit_records(cols_dev.data_ptr(), blob.data_ptr(),
                             FMT[record_type], 0, R, frame_off.data_ptr(),
                             file.data_ptr(), err.data_ptr(), _stream(),
                             total // max(R, 1))
    codes = torch.as_tensor(np.ascontiguousarray(part_codes, np.int64),
                            device=device)
    order = torch.argsort(codes, stable=True)
    sizes = psize[order].contiguous()
    dst_off = excl_sum(sizes)
    src_off = frame_off[:-1][order].contiguous()
    out = torch.empty(total, dtype=torch.uint8, device=device)
    _native.gpu_gather_payloads(file.data_ptr(), src_off.data_ptr(),
                                sizes.data_ptr(), dst_off.data_ptr(), R,
                                out.data_ptr(), _stream(), total // max(R, 1))
    # partition boundaries in the ordered space
    counts = torch.bincount(codes, minlength=num_parts)
    row_bound = torch.nn.functional.pad(torch.cumsum(counts, 0), (1, 0))
    byte_bound = dst_off[row_bound].cpu().numpy()
    _check_emit_err(err)
    ranges = [(p, int(byte_bound[p]), int(byte_bound[p + 1]))
              for p in range(num_parts) if byte_bound[p + 1] > byte_bound[p]]
    return out, ranges


# ---------------------------------------------------------------------------
# Schema inference on device: hash-table lattice kernel.
# ---------------------------------------------------------------------------

_INFER_SLOTS = 2048       # 1024 context + 1024 sequence feature names
_INFER_SLOTS_MAX = 1 << 21  # growth cap (1M distinct names per half)
_ERR_OVERFLOW = -5
_ERR_NAME_TOO_LONG = -6


def infer_codes_device(data: torch.Tensor, off: torch.Tensor,
                       lens: torch.Tensor, record_type: str) -> dict:
    """Device records -> {feature name: lattice code}. The kernel interns
    names into a device hash table and max-merges per-feature codes; the
    host only reads back the (tiny) table and resolves name strings. The
    table GROWS on overflow (datasets with thousands of distinct feature
    names rerun the kernel with 4x the slots; the scan is cheap relative to
    the decode it precedes, so the retry is simpler than spilling)."""
    check_native()
    device = data.device
    fmt = FMT["SequenceExample" if record_type == "SequenceExample" else "Example"]
    R = off.numel()
    slots = _INFER_SLOTS
    while True:
        table = torch.zeros((slots, 3), dtype=torch.int64, device=device)
        err = torch.zeros(2, dtype=torch.int32, device=device)
        if R:
            _native.gpu_infer_codes(data.data_ptr(), off.data_ptr(),
                                    lens.data_ptr(), R, fmt, table.data_ptr(),
                                    slots, err.data_ptr(), _stream())
        tab = table.cpu().numpy()
        rc = int(err[0].item())
        if rc == _ERR_OVERFLOW:
            if slots < _INFER_SLOTS_MAX:
                slots *= 4
                continue
            raise RuntimeError(
                "schema inference failed: more than "
                f"{_INFER_SLOTS_MAX // 2} distinct feature names")
        if rc == _ERR_NAME_TOO_LONG:
            raise RuntimeError(
                "schema inference failed: a feature name exceeds 65535 bytes")
        if rc != 0:
            rec = int(err[1].item()) - 1
            where = f" (record {rec})" if rec >= 0 else ""
            raise RuntimeError(f"malformed record during schema inference"
                               f"{where} (native error {rc})")
        break
    used = np.nonzero(tab[:, 0])[0]
    if len(used) == 0:
        return {}
    # recover name strings: one D2H of the covering range when it is small,
    # else one tiny D2H per distinct name (names are first occurrences and
    # can be anywhere in the file)
    refs = tab[used, 1]
    offs = (refs >> 16).astype(np.int64)
    nlens = (refs & 0xFFFF).astype(np.int64)
    lo = int(offs.min())
    hi = int((offs + nlens).max())
    codes: dict = {}
    if hi - lo <= (4 << 20):
        blob = data[lo:hi].cpu().numpy().tobytes()
        for o, ln, code in zip(offs, nlens, tab[used, 2]):
            name = blob[o - lo:o - lo + ln].decode("utf-8", errors="replace")
            codes[name] = max(codes.get(name, 0), int(code))
    else:
        for o, ln, code in zip(offs, nlens, tab[used, 2]):
            name = data[o:o + ln].cpu().numpy().tobytes().decode("utf-8", errors="replace")
            codes[name] = max(codes.get(name, 0), int(code))
    return codes


def infer_schema_file(path: str, record_type: str, device="cuda"):
    """Uncompressed file -> inferred StructType, fully on the GPU."""
    from ..infer import schema_from_codes

    data = read_file_to_device(path, device)
    off, lens = scan_frames_device(data)
    return schema_from_codes(infer_codes_device(data, off, lens, record_type))


def batch_to_device(batch: RecordBatch, device="cuda") -> RecordBatch:
    cols = []
    for c in batch.columns:
        cols.append(WireColumn(
            c.kind, c.is_seq,
            _dev(c.presence, torch.uint8, device),
            _dev(c.row_off, torch.int64, device),
            _dev(c.values, {KIND_INT64: torch.int64, KIND_FLOAT: torch.float32,
                            KIND_BYTES: torch.uint8}[c.kind], device),
            _dev(c.elem_off, torch.int64, device) if c.elem_off is not None else None,
            _dev(c.list_off, torch.int64, device) if c.list_off is not None else None,
            _dev(c.sub_off, torch.int64, device) if c.sub_off is not None else None,
        ))
    return RecordBatch(batch.schema, cols, batch.num_rows)


def batch_to_host(batch: RecordBatch) -> RecordBatch:
    """Device batch -> host numpy batch. All D2H copies go into PINNED
    host tensors (torch's caching host allocator reuses them) issued
    async round-robin over the TWO side streams with one sync at the end —
    pageable `.cpu()` per tensor runs at ~8 GB/s with a hidden staging
    copy, one pinned stream at ~35, two at link speed. The numpy arrays
    are zero-copy views of the pinned tensors (kept alive by the view's
    base)."""
    streams = _dma_streams()
    main = torch.cuda.current_stream()
    state = {"used": False, "k": 0}

    def h(t):
        if not isinstance(t, torch.Tensor):
            return t
        if not t.is_cuda:
            return t.numpy()
        host = torch.empty_like(t, device="cpu", pin_memory=True)
        if not state["used"]:
            for s in streams:
                s.wait_stream(main)
            state["used"] = True
        nbytes = t.numel() * t.element_size()
        if nbytes >= (8 << 20) and t.is_contiguous() and len(streams) > 1:
            # a single dominant tensor (ByteArray values, big value columns)
            # would ride ONE stream under round-robin and cap at the
            # single-blit rate (~21 GB/s measured); split it across both
            flat_d = t.view(-1)
            flat_h = host.view(-1)
            n = flat_d.numel()
            step = (n + len(streams) - 1) // len(streams)
            for i, st in enumerate(streams):
                lo = i * step
                hi = min(n, lo + step)
                if lo >= hi:
                    break
                with torch.cuda.stream(st):
                    flat_h[lo:hi].copy_(flat_d[lo:hi], non_blocking=True)
                t.record_stream(st)
            return host.numpy()
        st = streams[state["k"] % len(streams)]
        state["k"] += 1
        with torch.cuda.stream(st):
            host.copy_(t, non_blocking=True)
        t.record_stream(st)
        return host.numpy()

    cols = [WireColumn(c.kind, c.is_seq, h(c.presence), h(c.row_off), h(c.values),
                       h(c.elem_off) if c.elem_off is not None else None,
                       h(c.list_off) if c.list_off is not None else None,
                       h(c.sub_off) if c.sub_off is not None else None)
            for c in batch.columns]
    if state["used"]:
        for s in streams:
            s.synchronize()
    return RecordBatch(batch.schema, cols, batch.num_rows)


# ---------------------------------------------------------------------------
# CPU-boundary convenience wrappers (used by the file reader/writer)
# ---------------------------------------------------------------------------

def decode_buffer_to_cpu(data_np: np.ndarray, schema: StructType, record_type: str,
                         verify_crc: bool = True) -> RecordBatch:
    return batch_to_host(decode_buffer_device(data_np, schema, record_type,
                                              verify_crc))


def encode_batch_from_cpu(batch: RecordBatch, record_type: str) -> bytes:
    dev_batch = batch_to_device(batch)
    file = encode_device(dev_batch, record_type)
    return device_to_bytes(file)
