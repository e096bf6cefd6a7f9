"""CPU tests of the device DEFLATE inflater core (csrc/inflate_core.h).

The exact code that runs one-segment-per-lane on the GPU also compiles for
the host (`_native.host_inflate_segment`), so the algorithm is fully
testable here: numerics vs zlib across entropy profiles, block-type
coverage (stored / fixed / dynamic), malformed-stream rejection, and the
regression for the prefetch-refill underflow (a nearly-empty prefetch
register delivered <16 bits to a 16-bit read at segment tails)."""

import gzip
import zlib

import numpy as np
import pytest

from spark_tfrecord_amd import _native
from spark_tfrecord_amd.io import paths as P


def roundtrip(data: bytes):
    gz = P.compress_bytes(data, "gzip")
    meta = P.parse_gz_segments(gz)
    assert meta is not None
    body_off, segs, crc, isize = meta
    pos, upos = body_off, 0
    for i, (c, u) in enumerate(segs):
        out = _native.host_inflate_segment(gz[pos:pos + c], u)
        assert out == data[upos:upos + u], f"segment {i}"
        pos += c
        upos += u
    assert upos == len(data)


class TestInflateCore:
    def test_text_dynamic_huffman(self):
        rng = np.random.default_rng(0)
        roundtrip(bytes(rng.integers(65, 90, 1_500_000).astype(np.uint8)))

    def test_random_stored_blocks(self):
        roundtrip(np.random.default_rng(1).bytes(700_000))

    def test_highly_compressible(self):
        roundtrip((b"log line %08d with repeated content\n" * 20000)
                  % tuple(range(20000)))

    def test_tiny_and_empty(self):
        for d in (b"", b"a", b"ab" * 5, b"\x00" * 100):
            roundtrip(d)

    def test_exact_segment_boundaries(self):
        seg = P._GZ_SEGMENT
        rng = np.random.default_rng(2)
        for n in (seg - 1, seg, seg + 1, 2 * seg):
            roundtrip(bytes(rng.integers(97, 123, n).astype(np.uint8)))

    def test_fixed_huffman_blocks(self):
        # tiny compressible payloads make zlib pick FIXED codes
        c = zlib.compressobj(6, zlib.DEFLATED, -15)
        raw = c.compress(b"hellohellohello") + c.flush()
        out = _native.host_inflate_segment(raw, 15)
        assert out == b"hellohellohello"

    def test_malformed_streams_error_not_hang(self):
        rng = np.random.default_rng(3)
        data = bytes(rng.integers(65, 90, 200_000).astype(np.uint8))
        gz = P.compress_bytes(data, "gzip")
        body_off, segs, _, _ = P.parse_gz_segments(gz)
        c0, u0 = segs[0]
        seg = bytearray(gz[body_off:body_off + c0])
        for trial in range(60):
            bad = bytearray(seg)
            flips = rng.integers(0, len(bad), 3)
            for f in flips:
                bad[int(f)] ^= 1 << int(rng.integers(0, 8))
            try:
                out = _native.host_inflate_segment(bytes(bad), u0)
                # a flip may land in padding/ignored bits: output must
                # still be the exact expected length if it "succeeds"
                assert len(out) == u0
            except RuntimeError as e:
                assert "cause" in str(e)

    def test_truncated_stream_errors(self):
        data = b"x" * 50_000
        gz = P.compress_bytes(data, "gzip")
        body_off, segs, _, _ = P.parse_gz_segments(gz)
        c0, u0 = segs[0]
        seg = gz[body_off:body_off + c0]
        with pytest.raises(RuntimeError):
            _native.host_inflate_segment(seg[: len(seg) // 2], u0)

    @pytest.mark.parametrize("seed", range(5))
    def test_mixed_entropy_fuzz(self, seed):
        rng = np.random.default_rng(100 + seed)
        pieces = []
        for _ in range(int(rng.integers(3, 9))):
            n = int(rng.integers(1, 120_000))
            kind = int(rng.integers(0, 3))
            if kind == 0:
                pieces.append(rng.bytes(n))
            elif kind == 1:
                pieces.append(bytes(rng.integers(60, 70, n).astype(np.uint8)))
            else:
                pieces.append((b"abc123" * (n // 6 + 1))[:n])
        roundtrip(b"".join(pieces))


class TestRootTableEdges:
    """Root-table decode paths (10-bit lit, 8-bit dist tables with compare-
    chain fallback for longer codes): data shaped to force each regime."""

    def test_long_distance_codes_fallback(self):
        # far-apart repeats at many distances -> wide distance alphabet with
        # codes longer than the 8-bit dist table covers
        rng = np.random.default_rng(7)
        base = rng.bytes(40_000)
        parts = [base]
        for d in (33, 1025, 4097, 16385, 30000, 32767):
            parts.append(base[:200])
            parts.append(rng.bytes(d % 7000 + 100))
        parts.append(base)  # long match at large distance
        roundtrip(b"".join(parts))

    def test_skewed_literal_histogram_long_lit_codes(self):
        # extreme skew gives rare symbols 13-15 bit codes -> lit-table miss
        rng = np.random.default_rng(8)
        counts = np.ones(256, np.int64)
        counts[:4] = 200_000
        data = np.repeat(np.arange(256, dtype=np.uint8), counts)
        rng.shuffle(data[:800_000//2])
        roundtrip(bytes(data.tobytes()))

    def test_match_heavy_dictionary_text(self):
        words = [b"alpha", b"bravo", b"charlie", b"delta", b"echo", b"tfrec"]
        rng = np.random.default_rng(9)
        data = b" ".join(words[int(i)] for i in rng.integers(0, 6, 300_000))
        roundtrip(data)

    @pytest.mark.parametrize("seed", range(4))
    def test_bitflip_fuzz_tables(self, seed):
        """Corrupted streams must error, never hang or mis-decode silently
        through the table fast path."""
        rng = np.random.default_rng(100 + seed)
        data = b" ".join(
            [b"token%d" % i for i in rng.integers(0, 50, 20_000)])
        gz = P.compress_bytes(data, "gzip")
        meta = P.parse_gz_segments(gz)
        body_off, segs, crc, isize = meta
        c0, u0 = segs[0]
        seg = bytearray(gz[body_off:body_off + c0])
        for _ in range(60):
            i = int(rng.integers(0, len(seg)))
            bit = 1 << int(rng.integers(0, 8))
            seg[i] ^= bit
            try:
                out = _native.host_inflate_segment(bytes(seg), u0)
                # a flip may legitimately decode to different bytes; the
                # gzip layer catches that via CRC32. Here we only require
                # no hang/crash and correct length.
                assert len(out) == u0
            except RuntimeError:
                pass
            seg[i] ^= bit
