// DEFLATE (RFC 1951) segment inflater core, shared host/device.
//
// The gfx950 kernel (csrc/hip/inflate.hip) runs this one-segment-per-WAVE
// (64 lanes in lockstep on one bitstream — wave-uniform values, lane-split
// bulk copies, LDS scratch per wave); the host build backs the CPU tests
// (the algorithm is bit-exact on both) and a reference path for debugging.
// Design notes: docs/KERNELS.md "inflate_segments_kernel".
#pragma once

#include <cstdint>
#include <cstring>

#include "crc32c.h"  // TFR_HOSTDEV

// Wave-visibility fence for the cooperative copies (no-op on host; the
// builtin needs no runtime header, unlike __threadfence_block).
#if defined(__HIP_DEVICE_COMPILE__)
#define TFR_WAVE_FENCE() __builtin_amdgcn_fence(__ATOMIC_ACQ_REL, "workgroup")
#else
#define TFR_WAVE_FENCE() ((void)0)
#endif

namespace tfrec {
namespace inflate {

using u8 = uint8_t;
using u32 = uint32_t;
using u64 = uint64_t;
using i64 = int64_t;

// RFC 1951 length/distance decode tables (constant memory, shared by lanes).
constexpr uint16_t kLenBase[29] = {
    3,  4,  5,  6,  7,  8,  9,  10, 11,  13,  15,  17,  19,  23, 27,
    31, 35, 43, 51, 59, 67, 83, 99, 115, 131, 163, 195, 227, 258};
constexpr uint8_t kLenExtra[29] = {0, 0, 0, 0, 0, 0, 0, 0, 1, 1,
                                                 1, 1, 2, 2, 2, 2, 3, 3, 3, 3,
                                                 4, 4, 4, 4, 5, 5, 5, 5, 0};
constexpr uint16_t kDistBase[30] = {
    1,    2,    3,    4,    5,    7,     9,     13,    17,    25,
    33,   49,   65,   97,   129,  193,   257,   385,   513,   769,
    1025, 1537, 2049, 3073, 4097, 6145,  8193,  12289, 16385, 24577};
constexpr uint8_t kDistExtra[30] = {
    0, 0, 0, 0, 1, 1, 2, 2, 3, 3, 4,  4,  5,  5,  6,
    6, 7, 7, 8, 8, 9, 9, 10, 10, 11, 11, 12, 12, 13, 13};
constexpr uint8_t kClOrder[19] = {16, 17, 18, 0, 8,  7, 9,
                                                6,  10, 5,  11, 4, 12, 3,
                                                13, 2,  14, 1,  15};

// Decode scratch (~1 KB): ONE per wave on the GPU (all 64 lanes run the
// same segment in lockstep, so accesses are wave-uniform broadcasts), or
// plain stack state on the host. bc_* pack the watershed limit and the
// (rank - base) offset into one u32 per code length.
struct LaneScratch {
  u32 bc_lit[16];          // (cnt << 16) | first canonical code, per length
  u32 bc_dist[16];
  uint16_t sym[320];       // lit/len symbols [0,288) + dist symbols [288,320)
                           // (tail doubles as CL-table space during header)
  uint16_t rank_lit[17];   // first symbol-table rank per length
  uint16_t rank_dist[17];
  u8 lens4[160];           // 320 nibble-packed code lengths
};

TFR_HOSTDEV inline void set_len4(u8* a, int i, u32 v) {
  u8 m = a[i >> 1];
  a[i >> 1] = (i & 1) ? (u8)((m & 0x0F) | (v << 4)) : (u8)((m & 0xF0) | v);
}

TFR_HOSTDEV inline u32 get_len4(const u8* a, int i) {
  return (i & 1) ? (a[i >> 1] >> 4) : (a[i >> 1] & 0x0F);
}

struct BitRd {
  const u8* p;
  const u8* end;
  u64 buf;    // active bits
  int n;      // bit count; < 0 => underflow (sticky error)
  u64 pre;    // PREFETCHED next input bytes (low-order first)
  int pre_n;  // valid bytes in pre
};

// The next input word is loaded the moment the previous one is consumed —
// >=4 symbols of decode separate the load from its first use, hiding the
// global-load latency that otherwise sits on the serial decode chain
// (byte-at-a-time refills measured 15x slower; unprefetched word refills
// still stalled every ~4 symbols at the inflater's 1-wave/CU occupancy).
TFR_HOSTDEV inline void br_load_pre(BitRd& b) {
  if (b.end - b.p >= 8) {
    __builtin_memcpy(&b.pre, b.p, 8);
    b.p += 8;
    b.pre_n = 8;
  } else {
    b.pre = 0;
    b.pre_n = (int)(b.end - b.p);
    for (int i = 0; i < b.pre_n; ++i) b.pre |= (u64)b.p[i] << (8 * i);
    b.p = b.end;
  }
}

TFR_HOSTDEV inline void br_init(BitRd& b, const u8* in, i64 ilen) {
  b.p = in;
  b.end = in + ilen;
  b.buf = 0;
  b.n = 0;
  br_load_pre(b);
}

TFR_HOSTDEV inline void br_refill(BitRd& b) {
  // must leave n >= 57 unless the input is exhausted: a nearly-empty pre
  // can deliver as little as one byte, and br_bits(16) only refills once
  while (b.n <= 56) {
    if (b.pre_n == 0) {
      if (b.p >= b.end) break;
      br_load_pre(b);
    }
    int take = (64 - b.n) >> 3;
    if (take > b.pre_n) take = b.pre_n;
    b.buf |= b.pre << b.n;  // bits past 64 truncate; only `take` counted
    b.n += take << 3;
    b.pre = (take >= 8) ? 0 : (b.pre >> (take << 3));
    b.pre_n -= take;
  }
  if (b.pre_n == 0 && b.p < b.end) br_load_pre(b);  // prefetch the next word
}

TFR_HOSTDEV inline u32 br_bits(BitRd& b, int k) {
  if (b.n < k) {
    br_refill(b);
    if (b.n < k) {
      b.n = -(1 << 20);  // sticky underflow
      return 0;
    }
  }
  u32 v = (u32)(b.buf & ((1u << k) - 1u));
  b.buf >>= k;
  b.n -= k;
  return v;
}

// Canonical-Huffman build from nibble-packed lengths [len_off, len_off+nsym).
// Emits (bc = (cnt<<16)|base, rank, sym); rejects over-subscribed codes.
TFR_HOSTDEV inline bool build_huff4(const u8* lens4, int len_off, int nsym,
                                   u32* bc, uint16_t* rank, uint16_t* sym) {
  for (int l = 0; l < 17; ++l) rank[l] = 0;
  for (int s = 0; s < nsym; ++s) {
    u32 L = get_len4(lens4, len_off + s);
    if (L) ++rank[L];  // rank[] temporarily holds counts
  }
  u32 code = 0, k = 0;
  for (int l = 1; l <= 15; ++l) {
    u32 cnt = rank[l];
    // watershed form: hit at length l iff rev15 < lim (both 15-bit
    // justified); symbol = sym[(rev15 >> (15-l)) + offset] with offset =
    // rank - base in wraparound u16 arithmetic
    bc[l] = (((code + cnt) << (15 - l)) << 16) | (uint16_t)(k - code);
    rank[l] = (uint16_t)k;
    k += cnt;
    code = (code + cnt) << 1;
    if (code > (2u << l)) return false;  // over-subscribed
  }
  rank[16] = (uint16_t)k;
  uint16_t nxt[16];
  for (int l = 0; l < 16; ++l) nxt[l] = rank[l];
  for (int s = 0; s < nsym; ++s) {
    u32 L = get_len4(lens4, len_off + s);
    if (L) sym[nxt[L]++] = (uint16_t)s;
  }
  return true;
}

// Same build from a u16 length array (the 19-symbol code-length alphabet).
TFR_HOSTDEV inline bool build_huff16(const uint16_t* lens, int nsym,
                                    u32* bc, uint16_t* rank, uint16_t* sym) {
  for (int l = 0; l < 17; ++l) rank[l] = 0;
  for (int s = 0; s < nsym; ++s)
    if (lens[s]) ++rank[lens[s]];
  u32 code = 0, k = 0;
  for (int l = 1; l <= 15; ++l) {
    u32 cnt = rank[l];
    bc[l] = (((code + cnt) << (15 - l)) << 16) | (uint16_t)(k - code);
    rank[l] = (uint16_t)k;
    k += cnt;
    code = (code + cnt) << 1;
    if (code > (2u << l)) return false;
  }
  rank[16] = (uint16_t)k;
  uint16_t nxt[16];
  for (int l = 0; l < 16; ++l) nxt[l] = rank[l];
  for (int s = 0; s < nsym; ++s)
    if (lens[s]) sym[nxt[lens[s]]++] = (uint16_t)s;
  return true;
}

// Direct-lookup root table for the lit/len alphabet: 2^10 entries indexed
// by the next 10 bits (MSB-first, i.e. the high bits of the 15-bit
// reversed peek), each (sym << 4) | code_len for code lengths <= 10, 0 for
// longer/invalid prefixes. Collapses the per-symbol compare chain + sym[]
// load (~74 VALU + ~60 SALU per symbol measured by PMC — the whole decode
// runs redundantly on all 64 lanes, so instruction count IS the wall
// clock) to one LDS load + a few ALU ops. The fill is the one place the
// wave gets real SIMD parallelism: each lane derives 16 entries with the
// compare chain below, writing disjoint LDS slots.
constexpr int kLitTabBits = 10;
constexpr int kLitTabSize = 1 << kLitTabBits;
constexpr int kDistTabBits = 8;
constexpr int kDistTabSize = 1 << kDistTabBits;

template <int BITS>
TFR_HOSTDEV inline void fill_huff_table(const u32* bc, const uint16_t* sym,
                                        uint16_t* tab, int lane,
                                        int nlanes = 64) {
  int start = lane < 0 ? 0 : lane;
  int step = lane < 0 ? 1 : nlanes;
  for (int idx = start; idx < (1 << BITS); idx += step) {
    u32 rev = (u32)idx << (15 - BITS);  // left-justified prefix
    u32 e = 0;
    for (int l = 1; l <= BITS; ++l) {
      if (rev < (bc[l] >> 16)) {
        u32 s = sym[(uint16_t)((u32)(idx >> (BITS - l)) +
                               (u32)(uint16_t)bc[l])];
        e = (s << 4) | (u32)l;
        break;
      }
    }
    tab[idx] = (uint16_t)e;
  }
  TFR_WAVE_FENCE();  // all lanes read every lane's entries
}

// Table-first decode; falls back to the compare chain for code lengths
// > BITS (rare — 10 bits cover the whole fixed-code lit alphabet and
// virtually all dynamic-code literals; 8 bits cover typical distances).
template <int BITS>
TFR_HOSTDEV inline int huff_decode_tab(BitRd& br, const u32* bc,
                                       const uint16_t* sym,
                                       const uint16_t* tab) {
  if (br.n < 15) br_refill(br);  // short tail: zero bits pad the peek
  u32 rev = __builtin_bitreverse32((u32)br.buf) >> 17;
  u32 e = tab[rev >> (15 - BITS)];
  if (e) {
    int l = (int)(e & 15u);
    if (br.n < l) {
      br.n = -(1 << 20);  // consumed past the stream end
      return -1;
    }
    br.buf >>= l;
    br.n -= l;
    return (int)(e >> 4);
  }
  int l = 16;
  for (int k = BITS + 1; k <= 15; ++k) {
    if (rev < (bc[k] >> 16)) {
      l = k;
      break;
    }
  }
  if (l > 15 || br.n < l) {
    br.n = -(1 << 20);
    return -1;
  }
  u32 w = bc[l];
  br.buf >>= l;
  br.n -= l;
  return sym[(uint16_t)((rev >> (15 - l)) + (u32)(uint16_t)w)];
}

// Peek-based canonical decode: bit-reverse the next 15 buffered bits once,
// then every candidate length is a shift+compare. The (cnt|base) words for
// lengths 1..8 are INDEPENDENT of the bitstream, so they are batch-loaded
// up front — one LDS wait covers the whole common case instead of a
// dependent load per candidate length (the inflater runs at 1-2 waves per
// CU, so every serialized memory wait is raw wall time; PMC showed
// VALU/busy = 0.04 with the load-per-iteration form).
TFR_HOSTDEV inline int huff_decode(BitRd& br, const u32* bc,
                                  const uint16_t* rank, const uint16_t* sym) {
  (void)rank;  // folded into bc as (rank - base); kept for the builders
  if (br.n < 15) br_refill(br);  // short tail: zero bits pad the peek
  u32 rev = __builtin_bitreverse32((u32)br.buf) >> 17;
  u32 b1 = bc[1], b2 = bc[2], b3 = bc[3], b4 = bc[4];
  u32 b5 = bc[5], b6 = bc[6], b7 = bc[7], b8 = bc[8];
  int l;
  if (rev < (b1 >> 16)) l = 1;
  else if (rev < (b2 >> 16)) l = 2;
  else if (rev < (b3 >> 16)) l = 3;
  else if (rev < (b4 >> 16)) l = 4;
  else if (rev < (b5 >> 16)) l = 5;
  else if (rev < (b6 >> 16)) l = 6;
  else if (rev < (b7 >> 16)) l = 7;
  else if (rev < (b8 >> 16)) l = 8;
  else {
    l = 16;
    for (int k = 9; k <= 15; ++k) {
      if (rev < (bc[k] >> 16)) {
        l = k;
        break;
      }
    }
    if (l > 15) return -1;
  }
  if (br.n < l) {
    br.n = -(1 << 20);  // consumed past the stream end
    return -1;
  }
  u32 w = (l <= 4 ? (l <= 2 ? (l == 1 ? b1 : b2) : (l == 3 ? b3 : b4))
                  : (l <= 8 ? (l <= 6 ? (l == 5 ? b5 : b6) : (l == 7 ? b7 : b8))
                            : bc[l]));
  br.buf >>= l;
  br.n -= l;
  return sym[(uint16_t)((rev >> (15 - l)) + (u32)(uint16_t)w)];
}

// Bulk copy: serial on host (lane < 0); lane-strided u64s when `nlanes`
// lanes run ONE segment in lockstep (the GPU decomposition — per-lane
// segments paid a ~5x SIMT divergence tax on 64 independent bitstreams).
// Only used where src/dst cannot overlap within the stride window.
TFR_HOSTDEV inline void bulk_copy(u8* dst, const u8* src, i64 n, int lane,
                                  int nlanes = 64) {
  if (lane < 0) {
    i64 i = 0;
    for (; i + 8 <= n; i += 8) {
      u64 w;
      __builtin_memcpy(&w, src + i, 8);
      __builtin_memcpy(dst + i, &w, 8);
    }
    for (; i < n; ++i) dst[i] = src[i];
    return;
  }
  for (i64 i = (i64)lane * 8; i + 8 <= n; i += (i64)nlanes * 8) {
    u64 w;
    __builtin_memcpy(&w, src + i, 8);
    __builtin_memcpy(dst + i, &w, 8);
  }
  i64 tail = n & ~((i64)7);
  for (i64 b = tail + lane; b < n; b += nlanes) dst[b] = src[b];
  TFR_WAVE_FENCE();  // other lanes may read these bytes (later matches)
}

// Inflate one raw-deflate segment into dst[0, expect). Returns 0 on
// success, a small nonzero cause code otherwise (any nonzero => the Python
// side redoes the FILE on the host zlib path). `in`/`dst` are __restrict__:
// without it the compiler must order every bit-refill load after all
// pending output stores (they could alias), serializing the decode on L2
// store latency. `lane` < 0 = host/serial; otherwise the caller runs the
// WHOLE WAVE through this function in lockstep on one segment — every
// value is wave-uniform (the compiler keeps it in scalar registers), and
// `lane` only splits the big copies.
template <int NLANES = 64, int LIT_BITS = kLitTabBits,
          int DIST_BITS = kDistTabBits>
TFR_HOSTDEV inline int inflate_one(const u8* __restrict__ in, i64 ilen,
                                  u8* __restrict__ dst, i64 expect,
                                  LaneScratch& L, uint16_t* __restrict__ lit_tab,
                                  uint16_t* __restrict__ dist_tab,
                                  int lane = -1) {
  constexpr int nlanes = NLANES;
  BitRd br;
  br_init(br, in, ilen);
  i64 opos = 0;
  for (;;) {
    // a non-final segment ends after its full-flush empty stored block:
    // all output produced and fewer bits left than any block needs
    if (opos >= expect &&
        ((i64)(br.end - br.p) + br.pre_n) * 8 + br.n < 10)
      break;
    u32 final = br_bits(br, 1);
    u32 btype = br_bits(br, 2);
    if (br.n < 0) return 1;
    if (btype == 0) {  // stored
      br.buf >>= (br.n & 7);
      br.n &= ~7;
      u32 len = br_bits(br, 16);
      u32 nlen = br_bits(br, 16);
      if (br.n < 0 || ((len ^ nlen) & 0xFFFFu) != 0xFFFFu) return 2;
      // rewind both the bit buffer's and the prefetch register's bytes
      const u8* src = br.p - br.pre_n - (br.n >> 3);
      if (src + len > br.end || opos + (i64)len > expect) return 3;
      bulk_copy(dst + opos, src, (i64)len, lane, nlanes);
      opos += len;
      br.p = src + len;
      br.buf = 0;
      br.n = 0;
      br.pre = 0;
      br.pre_n = 0;
      br_load_pre(br);
      if (final) break;
      continue;
    }
    if (btype == 3) return 4;
    int hlit, hdist, dist_off;
    if (btype == 1) {  // fixed codes
      hlit = 288;
      hdist = 32;
      dist_off = 288;
      for (int s = 0; s < 144; ++s) set_len4(L.lens4, s, 8);
      for (int s = 144; s < 256; ++s) set_len4(L.lens4, s, 9);
      for (int s = 256; s < 280; ++s) set_len4(L.lens4, s, 7);
      for (int s = 280; s < 288; ++s) set_len4(L.lens4, s, 8);
      for (int s = 0; s < 32; ++s) set_len4(L.lens4, 288 + s, 5);
    } else {  // dynamic codes
      hlit = (int)br_bits(br, 5) + 257;
      hdist = (int)br_bits(br, 5) + 1;
      int hclen = (int)br_bits(br, 4) + 4;
      if (br.n < 0) return 5;
      dist_off = hlit;
      // CL table borrows the tail of sym[] (lit/dist fills happen later)
      uint16_t* cl_lens = &L.sym[300];  // 19 entries
      uint16_t* cl_sym = &L.sym[280];   // <= 19 entries
      for (int i = 0; i < 19; ++i) cl_lens[i] = 0;
      for (int i = 0; i < hclen; ++i) cl_lens[kClOrder[i]] = br_bits(br, 3);
      if (br.n < 0 || !build_huff16(cl_lens, 19, L.bc_dist, L.rank_dist,
                                    cl_sym))
        return 6;
      int total = hlit + hdist;
      int n = 0;
      u32 prev = 0;
      while (n < total) {
        int s = huff_decode(br, L.bc_dist, L.rank_dist, cl_sym);
        if (s < 0) return 7;
        if (s < 16) {
          set_len4(L.lens4, n++, (u32)s);
          prev = (u32)s;
        } else if (s == 16) {
          int r = 3 + (int)br_bits(br, 2);
          while (r-- && n < total) set_len4(L.lens4, n++, prev);
        } else if (s == 17) {
          int r = 3 + (int)br_bits(br, 3);
          while (r-- && n < total) set_len4(L.lens4, n++, 0);
        } else {
          int r = 11 + (int)br_bits(br, 7);
          while (r-- && n < total) set_len4(L.lens4, n++, 0);
        }
        if (br.n < 0) return 8;
      }
    }
    if (!build_huff4(L.lens4, 0, hlit, L.bc_lit, L.rank_lit, L.sym))
      return 9;
    if (!build_huff4(L.lens4, dist_off, hdist, L.bc_dist, L.rank_dist,
                     &L.sym[288]))
      return 10;
    fill_huff_table<LIT_BITS>(L.bc_lit, L.sym, lit_tab, lane, nlanes);
    fill_huff_table<DIST_BITS>(L.bc_dist, &L.sym[288], dist_tab, lane,
                               nlanes);
    // literal accumulation window: byte-per-literal global stores made the
    // literal-heavy path store-bound; 8 literals flush as one u64 store
    // (flushed before matches, which may read the freshly-written bytes)
    u64 lw = 0;
    int ln = 0;
    auto flush_lits = [&]() {
      if (!ln) return;
      if (opos + 8 <= expect) {
        // bytes past ln are garbage but lie before future output: they are
        // overwritten by construction (opos+8 <= expect)
        __builtin_memcpy(dst + opos, &lw, 8);
      } else {
        for (int i = 0; i < ln; ++i) dst[opos + i] = (u8)(lw >> (8 * i));
      }
      opos += ln;
      lw = 0;
      ln = 0;
    };
    for (;;) {
      int s = huff_decode_tab<LIT_BITS>(br, L.bc_lit, L.sym, lit_tab);
      if (s < 0) return 11;
      if (s < 256) {
        if (opos + ln >= expect) return 12;
        lw |= (u64)(u8)s << (8 * ln);
        if (++ln == 8) flush_lits();
      } else if (s == 256) {
        flush_lits();
        break;
      } else {
        flush_lits();
        s -= 257;
        if (s >= 29) return 13;
        i64 mlen = kLenBase[s] + (i64)br_bits(br, kLenExtra[s]);
        int d = huff_decode_tab<DIST_BITS>(br, L.bc_dist, &L.sym[288],
                                           dist_tab);
        if (d < 0 || d >= 30) return 14;
        i64 dist = kDistBase[d] + (i64)br_bits(br, kDistExtra[d]);
        if (br.n < 0) return 15;
        if (dist > opos || opos + mlen > expect) return 16;
        const u8* sp = dst + (opos - dist);
        u8* dp = dst + opos;
        if (lane >= 0 && dist >= mlen) {
          bulk_copy(dp, sp, mlen, lane, nlanes);  // non-overlap: strided
        } else if (dist >= 8) {
          i64 i = 0;
          for (; i + 8 <= mlen; i += 8) {
            u64 w;
            __builtin_memcpy(&w, sp + i, 8);
            __builtin_memcpy(dp + i, &w, 8);
          }
          for (; i < mlen; ++i) dp[i] = sp[i];
        } else {
          for (i64 i = 0; i < mlen; ++i) dp[i] = sp[i];
        }
        opos += mlen;
      }
    }
    if (final) break;
  }
  return (opos == expect) ? 0 : 17;
}


}  // namespace inflate
}  // namespace tfrec
