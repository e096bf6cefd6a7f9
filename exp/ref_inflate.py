"""Bit-exact Python port of csrc/inflate_core.h for debugging (host oracle)."""
import numpy as np

LENB = [3,4,5,6,7,8,9,10,11,13,15,17,19,23,27,31,35,43,51,59,67,83,99,115,131,163,195,227,258]
LENE = [0,0,0,0,0,0,0,0,1,1,1,1,2,2,2,2,3,3,3,3,4,4,4,4,5,5,5,5,0]
DISTB = [1,2,3,4,5,7,9,13,17,25,33,49,65,97,129,193,257,385,513,769,1025,1537,2049,3073,4097,6145,8193,12289,16385,24577]
DISTE = [0,0,0,0,1,1,2,2,3,3,4,4,5,5,6,6,7,7,8,8,9,9,10,10,11,11,12,12,13,13]
ORD = [16,17,18,0,8,7,9,6,10,5,11,4,12,3,13,2,14,1,15]
M64 = (1 << 64) - 1


def brev15(x):
    return int(format(x & 0x7FFF, "015b")[::-1], 2)


class BR:
    def __init__(s, b, trace=False):
        s.b = b; s.i = 0; s.buf = 0; s.n = 0; s.pre = 0; s.pre_n = 0
        s.trace = trace
        s.load_pre()

    def load_pre(s):
        take = min(8, len(s.b) - s.i)
        s.pre = int.from_bytes(s.b[s.i:s.i + take], "little")
        s.pre_n = take
        s.i += take

    def refill(s):
        while s.n <= 56:
            if s.pre_n == 0:
                if s.i >= len(s.b):
                    break
                s.load_pre()
            take = min((64 - s.n) >> 3, s.pre_n)
            s.buf |= (s.pre << s.n) & M64
            s.n += take << 3
            s.pre = 0 if take >= 8 else (s.pre >> (take << 3))
            s.pre_n -= take
        if s.pre_n == 0 and s.i < len(s.b):
            s.load_pre()

    def bits(s, k):
        if s.n < k:
            s.refill()
            if s.n < k:
                raise EOFError(f"underflow k={k} n={s.n} pre_n={s.pre_n} i={s.i}/{len(s.b)}")
        v = s.buf & ((1 << k) - 1)
        s.buf >>= k
        s.n -= k
        return v

    def remaining_bits(s):
        return (len(s.b) - s.i + s.pre_n) * 8 + s.n


def build(lens):
    cnt = [0] * 16
    for L in lens:
        if L:
            cnt[L] += 1
    bc = [0] * 16
    rank = [0] * 17
    code = 0
    k = 0
    for l in range(1, 16):
        bc[l] = ((((code + cnt[l]) << (15 - l)) & 0xFFFFFFFF) << 16) | ((k - code) & 0xFFFF)
        rank[l] = k
        k += cnt[l]
        code = (code + cnt[l]) << 1
        if code > (2 << l):
            raise ValueError("oversubscribed")
    nxt = list(rank[:16])
    sym = [0] * max(k, 1)
    for s_, L in enumerate(lens):
        if L:
            sym[nxt[L]] = s_
            nxt[L] += 1
    return bc, sym


def decode(br, bc, sym):
    if br.n < 15:
        br.refill()
    rev = brev15(br.buf & 0x7FFF)
    for l in range(1, 16):
        if rev < (bc[l] >> 16):
            if br.n < l:
                raise EOFError("decode past end")
            br.buf >>= l
            br.n -= l
            return sym[((rev >> (15 - l)) + (bc[l] & 0xFFFF)) & 0xFFFF]
    raise ValueError("bad code")


def inflate_one(seg, expect, trace=False):
    br = BR(seg, trace)
    out = bytearray()
    blocks = 0
    while True:
        if len(out) >= expect and br.remaining_bits() < 10:
            break
        fin = br.bits(1)
        btype = br.bits(2)
        blocks += 1
        if trace:
            print(f"block {blocks}: fin={fin} type={btype} out={len(out)} rem={br.remaining_bits()}")
        if btype == 0:
            br.buf >>= (br.n & 7)
            br.n &= ~7
            ln = br.bits(16)
            nl = br.bits(16)
            if (ln ^ nl) & 0xFFFF != 0xFFFF:
                raise ValueError(f"stored hdr blk{blocks} out={len(out)}")
            src = br.i - br.pre_n - (br.n >> 3)
            out += seg[src:src + ln]
            br.i = src + ln
            br.buf = 0
            br.n = 0
            br.pre_n = 0
            br.pre = 0
            br.load_pre()
            if fin:
                break
            continue
        if btype == 3:
            raise ValueError("btype 3")
        if btype == 1:
            lit_lens = [8] * 144 + [9] * 112 + [7] * 24 + [8] * 8
            dist_lens = [5] * 32
        else:
            hlit = br.bits(5) + 257
            hdist = br.bits(5) + 1
            hclen = br.bits(4) + 4
            cl = [0] * 19
            for j in range(hclen):
                cl[ORD[j]] = br.bits(3)
            cbc, csym = build(cl)
            lens = []
            while len(lens) < hlit + hdist:
                s_ = decode(br, cbc, csym)
                if s_ < 16:
                    lens.append(s_)
                elif s_ == 16:
                    lens += [lens[-1]] * (3 + br.bits(2))
                elif s_ == 17:
                    lens += [0] * (3 + br.bits(3))
                else:
                    lens += [0] * (11 + br.bits(7))
            lit_lens = lens[:hlit]
            dist_lens = lens[hlit:hlit + hdist]
        lbc, lsym = build(lit_lens)
        dbc, dsym = build(dist_lens)
        while True:
            s_ = decode(br, lbc, lsym)
            if s_ < 256:
                out.append(s_)
            elif s_ == 256:
                break
            else:
                s_ -= 257
                mlen = LENB[s_] + br.bits(LENE[s_])
                d = decode(br, dbc, dsym)
                dist = DISTB[d] + br.bits(DISTE[d])
                if dist > len(out):
                    raise ValueError("bad dist")
                for _ in range(mlen):
                    out.append(out[-dist])
        if fin:
            break
    return bytes(out)


if __name__ == "__main__":
    from spark_tfrecord_amd.io import paths as P
    rng = np.random.default_rng(0)
    data = bytes(rng.integers(65, 90, 3_000_000).astype(np.uint8))
    gz = P.compress_bytes(data, "gzip")
    body_off, segs, _, _ = P.parse_gz_segments(gz)
    pos = body_off
    upos = 0
    for i, (c, u) in enumerate(segs):
        seg = gz[pos:pos + c]
        truth = data[upos:upos + u]
        try:
            out = inflate_one(seg, u, trace=(i == 10))
            ok = out == truth
        except Exception as e:
            ok = False
            out = b""
            print(f"seg {i}: EXC {e}")
        if not ok:
            print(f"seg {i}: MISMATCH out={len(out)}")
            break
        pos += c
        upos += u
    else:
        print("all segments ok")
