// CRC32C (Castagnoli) + TFRecord masking, shared host/device.
//
// The TFRecord frame (SURVEY.md §1 "On-disk format"; behavior of the
// reference's external TFRecordWriter/TFRecordFileInputFormat, call sites
// TFRecordOutputWriter.scala:21,37 and TFRecordFileReader.scala:32,51):
//   uint64 length (LE) | uint32 masked_crc32c(length_bytes) |
//   byte data[length]  | uint32 masked_crc32c(data)
//   masked = ((crc >> 15) | (crc << 17)) + 0xa282ead8
//
// Implementation: slicing-by-8 with compile-time-generated tables so the same
// code runs on the x86 host and inside gfx950 kernels (tables land in .rodata
// on host and constant memory on device). The host additionally has an
// SSE4.2 hardware-CRC path.
#pragma once

#include <cstddef>
#include <cstdint>

#if defined(__HIPCC__)
#define TFR_HOSTDEV __host__ __device__
#else
#define TFR_HOSTDEV
#endif

namespace tfrec {

constexpr uint32_t kCrc32cPoly = 0x82F63B78u;  // reversed Castagnoli
constexpr uint32_t kMaskDelta = 0xA282EAD8u;

struct Crc32cTables {
  uint32_t t[8][256];
};

constexpr Crc32cTables make_crc32c_tables() {
  Crc32cTables tb{};
  for (uint32_t i = 0; i < 256; ++i) {
    uint32_t c = i;
    for (int k = 0; k < 8; ++k) c = (c & 1) ? (kCrc32cPoly ^ (c >> 1)) : (c >> 1);
    tb.t[0][i] = c;
  }
  for (uint32_t i = 0; i < 256; ++i) {
    uint32_t c = tb.t[0][i];
    for (int s = 1; s < 8; ++s) {
      c = tb.t[0][c & 0xFF] ^ (c >> 8);
      tb.t[s][i] = c;
    }
  }
  return tb;
}

inline constexpr Crc32cTables kCrcTables = make_crc32c_tables();

// Table-driven slicing-by-8. `tab` lets device code pass an LDS-staged copy.
TFR_HOSTDEV inline uint32_t crc32c_sw(const uint8_t* p, size_t n,
                                      uint32_t crc = 0,
                                      const uint32_t (*tab)[256] = kCrcTables.t) {
  crc = ~crc;
  // Align to 8 bytes.
  while (n && (reinterpret_cast<uintptr_t>(p) & 7u)) {
    crc = tab[0][(crc ^ *p++) & 0xFF] ^ (crc >> 8);
    --n;
  }
  while (n >= 8) {
    uint64_t w;
    __builtin_memcpy(&w, p, 8);
    w ^= crc;  // little-endian
    crc = tab[7][w & 0xFF] ^ tab[6][(w >> 8) & 0xFF] ^ tab[5][(w >> 16) & 0xFF] ^
          tab[4][(w >> 24) & 0xFF] ^ tab[3][(w >> 32) & 0xFF] ^
          tab[2][(w >> 40) & 0xFF] ^ tab[1][(w >> 48) & 0xFF] ^
          tab[0][(w >> 56) & 0xFF];
    p += 8;
    n -= 8;
  }
  while (n--) crc = tab[0][(crc ^ *p++) & 0xFF] ^ (crc >> 8);
  return ~crc;
}

#if defined(__SSE4_2__) && !defined(__HIP_DEVICE_COMPILE__)
inline uint32_t crc32c_hw(const uint8_t* p, size_t n, uint32_t crc = 0) {
  crc = ~crc;
  while (n && (reinterpret_cast<uintptr_t>(p) & 7u)) {
    crc = __builtin_ia32_crc32qi(crc, *p++);
    --n;
  }
  uint64_t c64 = crc;
  while (n >= 8) {
    uint64_t w;
    __builtin_memcpy(&w, p, 8);
    c64 = __builtin_ia32_crc32di(c64, w);
    p += 8;
    n -= 8;
  }
  crc = static_cast<uint32_t>(c64);
  while (n--) crc = __builtin_ia32_crc32qi(crc, *p++);
  return ~crc;
}
#endif

TFR_HOSTDEV inline uint32_t crc32c(const uint8_t* p, size_t n, uint32_t crc = 0) {
#if defined(__SSE4_2__) && !defined(__HIP_DEVICE_COMPILE__)
  return crc32c_hw(p, n, crc);
#else
  return crc32c_sw(p, n, crc);
#endif
}

TFR_HOSTDEV inline uint32_t mask_crc(uint32_t crc) {
  return ((crc >> 15) | (crc << 17)) + kMaskDelta;
}

// ---------------------------------------------------------------------------
// CRC32C combination (zlib crc32_combine adapted to the Castagnoli
// polynomial): crc(A||B) from crc(A), crc(B) and len(B). CRC shifting by
// len(B) zero bytes is a linear operator over GF(2); it is applied with
// 32x32 bit-matrices built by squaring. Enables wavefront-cooperative CRC:
// 64 lanes each CRC a contiguous chunk, then 6 shuffle-combine levels.
// ---------------------------------------------------------------------------

struct CrcMat {
  uint32_t m[32];  // column i = operator applied to the unit vector 1<<i
};

TFR_HOSTDEV inline uint32_t crcmat_times(const CrcMat& a, uint32_t vec) {
  uint32_t sum = 0;
  int i = 0;
  while (vec) {
    if (vec & 1u) sum ^= a.m[i];
    vec >>= 1;
    ++i;
  }
  return sum;
}

TFR_HOSTDEV inline void crcmat_square(CrcMat& out, const CrcMat& a) {
  for (int i = 0; i < 32; ++i) out.m[i] = crcmat_times(a, a.m[i]);
}

// Builds the operator that advances a finalized CRC32C past `len` zero
// bytes (i.e. M8^len where M8 shifts one byte). Fold uses: combined =
// op * crc_left ^ crc_right. Building costs ~log2(len) matrix squarings —
// callers folding many equal-length segments build it ONCE.
TFR_HOSTDEV inline void crc32c_shift_op(CrcMat& out, uint64_t len) {
  CrcMat even, odd;
  odd.m[0] = kCrc32cPoly;  // operator for one zero BIT
  for (int i = 1; i < 32; ++i) odd.m[i] = 1u << (i - 1);
  crcmat_square(even, odd);  // 2 bits
  crcmat_square(odd, even);  // 4 bits
  crcmat_square(even, odd);  // 8 bits == one zero byte
  // out = identity
  for (int i = 0; i < 32; ++i) out.m[i] = 1u << i;
  CrcMat* cur = &even;
  CrcMat* nxt = &odd;
  uint64_t n = len;
  while (n) {
    if (n & 1u) {
      CrcMat tmp;
      for (int i = 0; i < 32; ++i) tmp.m[i] = crcmat_times(*cur, out.m[i]);
      out = tmp;
    }
    n >>= 1;
    if (!n) break;
    crcmat_square(*nxt, *cur);
    CrcMat* t = cur;
    cur = nxt;
    nxt = t;
  }
}

// crc(A||B) given crc of A (unmasked, finalized), crc of B, and len(B).
TFR_HOSTDEV inline uint32_t crc32c_combine(uint32_t crc1, uint32_t crc2,
                                           uint64_t len2) {
  if (len2 == 0) return crc1;
  CrcMat op;
  crc32c_shift_op(op, len2);
  return crcmat_times(op, crc1) ^ crc2;
}

// ---------------------------------------------------------------------------
// Register-only CRC shifting via GF(2^32) field multiplication. The matrix
// form above keeps 128-byte CrcMat arrays with runtime-indexed accesses —
// on the GPU those live in SCRATCH memory, and building one per record made
// the wave-CRC kernels 30x slower than the CRC work itself (measured:
// 15.6 ms vs 0.5 ms on the 16 KB-record decode). This form is pure
// registers: a CRC word w represents the polynomial sum_i w_i x^(31-i) in
// GF(2)[x]/P(x); advancing a CRC past L zero bytes is multiplication by
// alpha^(8L) where alpha = x, and alpha^(2^k) is a compile-time table.
// ---------------------------------------------------------------------------

// One zero BIT: state *= x (the classic reflected LFSR step).
TFR_HOSTDEV constexpr uint32_t crc_mulx(uint32_t c) {
  return (c >> 1) ^ (kCrc32cPoly & (0u - (c & 1u)));
}

// Field multiply in the CRC representation (bit 31 = x^0 = the identity).
TFR_HOSTDEV constexpr uint32_t crc_gfmul(uint32_t a, uint32_t b) {
  uint32_t res = 0;
  uint32_t cur = a;
  for (int i = 31; i >= 0; --i) {  // bit i of b is the alpha^(31-i) term
    if ((b >> i) & 1u) res ^= cur;
    cur = crc_mulx(cur);
  }
  return res;
}

struct CrcPow2 {
  uint32_t p[64];  // p[k] = alpha^(2^k)
};

constexpr CrcPow2 make_crc_pow2() {
  CrcPow2 t{};
  t.p[0] = crc_mulx(0x80000000u);  // alpha^1
  for (int k = 1; k < 64; ++k) t.p[k] = crc_gfmul(t.p[k - 1], t.p[k - 1]);
  return t;
}

inline constexpr CrcPow2 kCrcPow2 = make_crc_pow2();

// alpha^(8*nbytes): the field element whose product shifts a CRC past
// `nbytes` zero bytes. ~popcount(8*nbytes) gf multiplies, all registers.
TFR_HOSTDEV inline uint32_t crc32c_shift_elem(uint64_t nbytes) {
  uint64_t e = nbytes << 3;  // bit count
  uint32_t t = 0x80000000u;  // identity
  int k = 0;
  while (e) {
    if (e & 1u) t = crc_gfmul(t, kCrcPow2.p[k]);
    e >>= 1;
    ++k;
  }
  return t;
}

// crc(A||B) via the field form; identical result to crc32c_combine.
TFR_HOSTDEV inline uint32_t crc32c_combine_fast(uint32_t crc1, uint32_t crc2,
                                                uint64_t len2) {
  if (len2 == 0) return crc1;
  return crc_gfmul(crc1, crc32c_shift_elem(len2)) ^ crc2;
}

TFR_HOSTDEV inline uint32_t unmask_crc(uint32_t masked) {
  uint32_t rot = masked - kMaskDelta;
  return (rot << 15) | (rot >> 17);
}

TFR_HOSTDEV inline uint32_t masked_crc32c(const uint8_t* p, size_t n) {
  return mask_crc(crc32c(p, n));
}

}  // namespace tfrec
