// Shared host/device protobuf codec core for tf.Example / tf.SequenceExample.
//
// This is a from-scratch, schema-driven implementation of exactly the proto
// subset the TFRecord formats use (SURVEY.md §1 "On-disk format"):
//   Example           { Features features = 1; }
//   Features          { map<string, Feature> feature = 1; }
//   Feature           { oneof: BytesList=1 | FloatList=2 | Int64List=3 }
//   BytesList         { repeated bytes value = 1; }
//   FloatList         { repeated float value = 1 [packed]; }
//   Int64List         { repeated int64 value = 1 [packed]; }
//   SequenceExample   { Features context = 1; FeatureLists feature_lists = 2; }
//   FeatureLists      { map<string, FeatureList> feature_list = 1; }
//   FeatureList       { repeated Feature feature = 1; }
//
// All functions are __host__ __device__ so the identical logic backs the CPU
// path (csrc/host_codec.cpp) and the gfx950 kernels (csrc/hip/*.hip): records
// are the parallel axis on the GPU, bytes within a record are sequential here.
//
// Columnar ("wire-form") layout shared with Python (spark_tfrecord_amd/columnar.py):
//   non-seq field : presence u8[R]; row_off i64[R+1] (values per row, cumulative)
//   seq field     : + list_off i64[R+1] (sub-lists per row), sub_off i64[L+1]
//                   (values per sub-list); row_off then covers values per ROW
//   bytes kind    : values = flat u8; elem_off i64[E+1] (bytes per string)
//   int64 / float : values = i64[V] / f32[V]
#pragma once

#include <cstdint>
#include <cstring>

#include "crc32c.h"

namespace tfrec {

using i64 = int64_t;
using u64 = uint64_t;
using u32 = uint32_t;
using u8 = uint8_t;

// Feature oneof field numbers double as kind codes (schema.py KIND_*).
enum Kind : int32_t { KIND_BYTES = 1, KIND_FLOAT = 2, KIND_INT64 = 3 };

enum RecordFormat : int32_t { FMT_EXAMPLE = 0, FMT_SEQUENCE = 1, FMT_BYTE_ARRAY = 2 };

// Error codes (negative) returned by parse/size routines.
enum CodecErr : int32_t {
  ERR_OK = 0,
  ERR_TRUNCATED = -1,
  ERR_BAD_VARINT = -2,
  ERR_KIND_MISMATCH = -3,
  ERR_BAD_WIRETYPE = -4,
  ERR_OVERFLOW = -5,
  // feature name longer than the 16-bit length the inference kernel's
  // nameref packing carries (the reference has no such names in practice;
  // erroring beats silently corrupting the ref)
  ERR_NAME_TOO_LONG = -6,
};

// ---------------------------------------------------------------------------
// varint
// ---------------------------------------------------------------------------

// clang lowers these builtins natively for both host and gfx950 device code
TFR_HOSTDEV inline int popcount64(u64 x) { return __builtin_popcountll(x); }

TFR_HOSTDEV inline int ctz64(u64 x) { return __builtin_ctzll(x); }  // x != 0

constexpr u64 kMsbMask = 0x8080808080808080ull;

// Reads a base-128 varint; returns new cursor or nullptr on malformed/overrun.
// Fast path: ONE unaligned 8-byte load covers varints up to 8 bytes (the
// continuation-bit scan finds the terminator in registers) — the per-byte
// dependent-load loop only runs at buffer tails and for 9/10-byte varints.
// gfx950 supports unaligned global loads, so the memcpy lowers to a plain
// dwordx2 load; this is the hot instruction of the decode structure scan.
TFR_HOSTDEV inline const u8* read_varint(const u8* p, const u8* end, u64* out) {
  if (end - p >= 8) {
    u64 w;
    __builtin_memcpy(&w, p, 8);
    u64 stops = ~w & kMsbMask;
    if (stops) {
      int nb = (ctz64(stops) >> 3) + 1;
      u64 v = 0;
      for (int i = 0; i < nb; ++i)
        v |= ((w >> (8 * i)) & 0x7F) << (7 * i);
      *out = v;
      return p + nb;
    }
    u64 v = 0;
    for (int i = 0; i < 8; ++i) v |= ((w >> (8 * i)) & 0x7F) << (7 * i);
    const u8* q = p + 8;
    for (int shift = 56; q < end && shift < 64; shift += 7) {
      u8 b = *q++;
      v |= static_cast<u64>(b & 0x7F) << shift;
      if (!(b & 0x80)) {
        *out = v;
        return q;
      }
    }
    return nullptr;
  }
  u64 v = 0;
  int shift = 0;
  while (p < end) {
    u8 b = *p++;
    v |= static_cast<u64>(b & 0x7F) << shift;
    if (!(b & 0x80)) {
      *out = v;
      return p;
    }
    shift += 7;
    if (shift >= 64) return nullptr;
  }
  return nullptr;
}

TFR_HOSTDEV inline int varint_size(u64 v) {
  int n = 1;
  while (v >= 0x80) {
    v >>= 7;
    ++n;
  }
  return n;
}

TFR_HOSTDEV inline u8* write_varint(u8* p, u64 v) {
  while (v >= 0x80) {
    *p++ = static_cast<u8>(v) | 0x80;
    v >>= 7;
  }
  *p++ = static_cast<u8>(v);
  return p;
}

// ---------------------------------------------------------------------------
// Schema blob: [i32 nfields][FieldDescRaw x n][name bytes]
// Built in Python (columnar.py), identical bytes shipped to host calls and
// device constant buffers.
// ---------------------------------------------------------------------------

struct FieldDescRaw {
  int32_t kind;      // Kind
  int32_t is_seq;    // 1 => SequenceExample feature_lists entry (2-D ragged)
  int32_t name_off;  // offset into the names section
  int32_t name_len;
};

struct SchemaView {
  int32_t nfields;
  const FieldDescRaw* fields;
  const u8* names;

  TFR_HOSTDEV const u8* name(int f) const { return names + fields[f].name_off; }
  TFR_HOSTDEV int name_len(int f) const { return fields[f].name_len; }
};

TFR_HOSTDEV inline SchemaView schema_view(const u8* blob) {
  SchemaView v;
  int32_t n;
  __builtin_memcpy(&n, blob, 4);
  v.nfields = n;
  v.fields = reinterpret_cast<const FieldDescRaw*>(blob + 4);
  v.names = blob + 4 + n * static_cast<int>(sizeof(FieldDescRaw));
  return v;
}

TFR_HOSTDEV inline bool name_eq(const u8* a, const u8* b, int n) {
  while (n >= 8) {  // unaligned word compares (hot in the per-entry map walk)
    u64 x, y;
    __builtin_memcpy(&x, a, 8);
    __builtin_memcpy(&y, b, 8);
    if (x != y) return false;
    a += 8;
    b += 8;
    n -= 8;
  }
  for (int i = 0; i < n; ++i)
    if (a[i] != b[i]) return false;
  return true;
}

// Linear schema lookup by feature name. Schemas are tens of fields; records
// are the parallel axis, so this inner scan is cheap and branch-uniform.
TFR_HOSTDEV inline int schema_find(const SchemaView& s, const u8* name, int len,
                                   int want_seq) {
  for (int f = 0; f < s.nfields; ++f) {
    if (s.fields[f].is_seq == want_seq && s.fields[f].name_len == len &&
        name_eq(s.name(f), name, len))
      return f;
  }
  return -1;
}

// ---------------------------------------------------------------------------
// Decode pass A: per-record structure scan.
// For each schema field, locate the Feature (or FeatureList) body and count
// elements. Mirrors what the reference's deserializer dispatch does per row
// (TFRecordDeserializer.scala:21-61) but split into stats so the GPU can
// prefix-sum offsets between passes.
// ---------------------------------------------------------------------------

struct FieldStat {
  i64 pos;         // absolute byte offset of the body within the data buffer; -1 absent
  i64 len;         // body length in bytes
  i64 nvals;       // total scalar elements (ints / floats / strings)
  i64 nbytes;      // total string payload bytes (bytes kind only)
  i64 nlists;      // sub-list count (seq fields only)
  int32_t kind_found;  // observed oneof kind, 0 if absent/empty Feature
  int32_t err;
};

TFR_HOSTDEV inline void field_stat_clear(FieldStat* st) {
  st->pos = -1;
  st->len = 0;
  st->nvals = 0;
  st->nbytes = 0;
  st->nlists = 0;
  st->kind_found = 0;
  st->err = 0;
}

// Skips one field value of the given wire type. Returns new cursor or nullptr.
TFR_HOSTDEV inline const u8* skip_field(const u8* p, const u8* end, u32 wiretype) {
  u64 tmp;
  switch (wiretype) {
    case 0:  // varint
      return read_varint(p, end, &tmp);
    case 1:  // fixed64
      return (end - p >= 8) ? p + 8 : nullptr;
    case 2:  // length-delimited
      p = read_varint(p, end, &tmp);
      if (!p || static_cast<u64>(end - p) < tmp) return nullptr;
      return p + tmp;
    case 5:  // fixed32
      return (end - p >= 4) ? p + 4 : nullptr;
    default:
      return nullptr;
  }
}

// Scans one list submessage body (BytesList/FloatList/Int64List content) and
// counts elements. Accepts both packed and unpacked encodings for the numeric
// kinds (parsers must; TF itself writes packed).
TFR_HOSTDEV inline int32_t count_list_body(const u8* p, const u8* end, int32_t kind,
                                           i64* nvals, i64* nbytes) {
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) return ERR_BAD_VARINT;
    u32 fieldno = static_cast<u32>(tag >> 3);
    u32 wt = static_cast<u32>(tag & 7);
    if (fieldno != 1) {  // unknown field inside the list message: skip
      p = skip_field(p, end, wt);
      if (!p) return ERR_TRUNCATED;
      continue;
    }
    if (kind == KIND_BYTES) {
      if (wt != 2) return ERR_BAD_WIRETYPE;
      u64 len;
      p = read_varint(p, end, &len);
      if (!p || static_cast<u64>(end - p) < len) return ERR_TRUNCATED;
      *nvals += 1;
      *nbytes += static_cast<i64>(len);
      p += len;
    } else if (kind == KIND_FLOAT) {
      if (wt == 2) {  // packed
        u64 len;
        p = read_varint(p, end, &len);
        if (!p || static_cast<u64>(end - p) < len) return ERR_TRUNCATED;
        *nvals += static_cast<i64>(len / 4);
        p += len;
      } else if (wt == 5) {
        if (end - p < 4) return ERR_TRUNCATED;
        *nvals += 1;
        p += 4;
      } else {
        return ERR_BAD_WIRETYPE;
      }
    } else {  // KIND_INT64
      if (wt == 2) {  // packed varints
        u64 len;
        p = read_varint(p, end, &len);
        if (!p || static_cast<u64>(end - p) < len) return ERR_TRUNCATED;
        const u8* q = p;
        const u8* qe = p + len;
        // count = number of terminator bytes (MSB clear), found by popcount
        // over 8-byte windows — no per-varint parsing in the count pass
        if (len && (qe[-1] & 0x80)) return ERR_BAD_VARINT;  // ends mid-varint
        i64 cnt = 0;
        while (qe - q >= 8) {
          u64 w;
          __builtin_memcpy(&w, q, 8);
          cnt += popcount64(~w & kMsbMask);
          q += 8;
        }
        for (; q < qe; ++q) cnt += !(*q & 0x80);
        *nvals += cnt;
        p = qe;
      } else if (wt == 0) {
        u64 v;
        p = read_varint(p, end, &v);
        if (!p) return ERR_BAD_VARINT;
        *nvals += 1;
      } else {
        return ERR_BAD_WIRETYPE;
      }
    }
  }
  return ERR_OK;
}

// Scans one Feature message body: identifies the oneof kind and counts
// elements. `expect_kind` < 0 means "report what you find" (inference);
// otherwise a different kind is ERR_KIND_MISMATCH (reference behavior:
// TFRecordDeserializer.scala:177-221 kind checks).
TFR_HOSTDEV inline int32_t scan_feature_body(const u8* p, const u8* end,
                                             int32_t expect_kind, int32_t* kind_found,
                                             i64* nvals, i64* nbytes) {
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) return ERR_BAD_VARINT;
    u32 fieldno = static_cast<u32>(tag >> 3);
    u32 wt = static_cast<u32>(tag & 7);
    if (fieldno >= 1 && fieldno <= 3 && wt == 2) {
      u64 len;
      p = read_varint(p, end, &len);
      if (!p || static_cast<u64>(end - p) < len) return ERR_TRUNCATED;
      *kind_found = static_cast<int32_t>(fieldno);
      if (expect_kind >= 0 && static_cast<int32_t>(fieldno) != expect_kind)
        return ERR_KIND_MISMATCH;
      int32_t rc = count_list_body(p, p + len, static_cast<int32_t>(fieldno), nvals,
                                   nbytes);
      if (rc != ERR_OK) return rc;
      p += len;
    } else {
      p = skip_field(p, end, wt);
      if (!p) return ERR_TRUNCATED;
    }
  }
  return ERR_OK;
}

// Scans a Features message body (map<string, Feature>) and fills stats for
// schema-matched entries. `base` is the absolute offset of `p` within the
// file buffer so stats can store absolute positions.
TFR_HOSTDEV inline int32_t scan_features_body(const u8* p, const u8* end, i64 base,
                                              const SchemaView& schema, int want_seq,
                                              FieldStat* stats) {
  const u8* body_start = p;
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) return ERR_BAD_VARINT;
    u32 fieldno = static_cast<u32>(tag >> 3);
    u32 wt = static_cast<u32>(tag & 7);
    if (fieldno != 1 || wt != 2) {
      p = skip_field(p, end, wt);
      if (!p) return ERR_TRUNCATED;
      continue;
    }
    u64 entry_len;
    p = read_varint(p, end, &entry_len);
    if (!p || static_cast<u64>(end - p) < entry_len) return ERR_TRUNCATED;
    const u8* ep = p;
    const u8* ee = p + entry_len;
    p = ee;
    // Map entry: key = 1 (string), value = 2 (Feature / FeatureList message).
    const u8* key = nullptr;
    u64 key_len = 0;
    const u8* val = nullptr;
    u64 val_len = 0;
    while (ep < ee) {
      u64 etag;
      ep = read_varint(ep, ee, &etag);
      if (!ep) return ERR_BAD_VARINT;
      u32 efn = static_cast<u32>(etag >> 3);
      u32 ewt = static_cast<u32>(etag & 7);
      if (efn == 1 && ewt == 2) {
        ep = read_varint(ep, ee, &key_len);
        if (!ep || static_cast<u64>(ee - ep) < key_len) return ERR_TRUNCATED;
        key = ep;
        ep += key_len;
      } else if (efn == 2 && ewt == 2) {
        ep = read_varint(ep, ee, &val_len);
        if (!ep || static_cast<u64>(ee - ep) < val_len) return ERR_TRUNCATED;
        val = ep;
        ep += val_len;
      } else {
        ep = skip_field(ep, ee, ewt);
        if (!ep) return ERR_TRUNCATED;
      }
    }
    if (!key) continue;
    int f = schema_find(schema, key, static_cast<int>(key_len), want_seq);
    if (f < 0) continue;  // unknown feature: ignored, like the reference
    FieldStat* st = &stats[f];
    // duplicate map keys are valid protobuf with LAST-entry-wins semantics
    // (what protobuf-java gives the reference); reset the accumulators so
    // only the final body's counts survive — otherwise the prefix sums
    // reserve slots the extract pass never fills (garbage in the column)
    st->nvals = 0;
    st->nbytes = 0;
    st->nlists = 0;
    st->kind_found = 0;
    st->err = 0;
    st->pos = base + (val ? (val - body_start) : 0);
    st->len = static_cast<i64>(val_len);
    if (!want_seq) {
      int32_t rc = scan_feature_body(val, val + val_len, schema.fields[f].kind,
                                     &st->kind_found, &st->nvals, &st->nbytes);
      if (rc != ERR_OK) st->err = rc;
    } else {
      // FeatureList body: repeated Feature (field 1).
      const u8* lp = val;
      const u8* le = val + val_len;
      while (lp < le) {
        u64 ltag;
        lp = read_varint(lp, le, &ltag);
        if (!lp) {
          st->err = ERR_BAD_VARINT;
          break;
        }
        u32 lfn = static_cast<u32>(ltag >> 3);
        u32 lwt = static_cast<u32>(ltag & 7);
        if (lfn == 1 && lwt == 2) {
          u64 flen;
          lp = read_varint(lp, le, &flen);
          if (!lp || static_cast<u64>(le - lp) < flen) {
            st->err = ERR_TRUNCATED;
            break;
          }
          st->nlists += 1;
          int32_t kf = 0;
          int32_t rc = scan_feature_body(lp, lp + flen, schema.fields[f].kind, &kf,
                                         &st->nvals, &st->nbytes);
          if (rc != ERR_OK) {
            st->err = rc;
            break;
          }
          if (kf) st->kind_found = kf;
          lp += flen;
        } else {
          lp = skip_field(lp, le, lwt);
          if (!lp) {
            st->err = ERR_TRUNCATED;
            break;
          }
        }
      }
    }
  }
  return ERR_OK;
}

// Pass A entry point: scan one record payload. `fmt` selects Example vs
// SequenceExample framing of the top-level message. `abs_off` is the record's
// absolute offset in the data buffer. stats[f] must be pre-cleared.
TFR_HOSTDEV inline int32_t scan_record(const u8* data, i64 abs_off, i64 rec_len,
                                       int32_t fmt, const SchemaView& schema,
                                       FieldStat* stats) {
  const u8* p = data + abs_off;
  const u8* end = p + rec_len;
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) return ERR_BAD_VARINT;
    u32 fieldno = static_cast<u32>(tag >> 3);
    u32 wt = static_cast<u32>(tag & 7);
    bool is_features =
        (fmt == FMT_EXAMPLE && fieldno == 1) || (fmt == FMT_SEQUENCE && fieldno == 1);
    bool is_feature_lists = (fmt == FMT_SEQUENCE && fieldno == 2);
    if ((is_features || is_feature_lists) && wt == 2) {
      u64 len;
      p = read_varint(p, end, &len);
      if (!p || static_cast<u64>(end - p) < len) return ERR_TRUNCATED;
      i64 base = (p - data);
      if (is_features) {
        int32_t rc = scan_features_body(p, p + len, base, schema, 0, stats);
        if (rc != ERR_OK) return rc;
      } else {
        int32_t rc = scan_features_body(p, p + len, base, schema, 1, stats);
        if (rc != ERR_OK) return rc;
      }
      p += len;
    } else {
      p = skip_field(p, end, wt);
      if (!p) return ERR_TRUNCATED;
    }
  }
  return ERR_OK;
}

// ---------------------------------------------------------------------------
// Fused single-pass structure scan + payload CRC32C (cursor form).
// The split form loads every payload byte at least twice: once through the
// varint parser and once through the CRC loop. The cursor walks the record
// ONCE in 8-byte windows; each window feeds the parser (funnel-shifted
// peeks) and, when fully consumed, one slicing-by-8 CRC step. Key bytes are
// additionally read directly for the schema-name compare (short).
// Records whose map entries put the value before the key (legal protobuf,
// never produced by TF or this library) return ERR_RETRY_UNFUSED and the
// caller re-scans them with the two-pass form.
// ---------------------------------------------------------------------------

constexpr int32_t ERR_RETRY_UNFUSED = -100;

TFR_HOSTDEV inline u32 crc32c_step8(u32 crc, u64 w, const uint32_t (*tab)[256]) {
  w ^= crc;
  return tab[7][w & 0xFF] ^ tab[6][(w >> 8) & 0xFF] ^ tab[5][(w >> 16) & 0xFF] ^
         tab[4][(w >> 24) & 0xFF] ^ tab[3][(w >> 32) & 0xFF] ^
         tab[2][(w >> 40) & 0xFF] ^ tab[1][(w >> 48) & 0xFF] ^
         tab[0][(w >> 56) & 0xFF];
}

struct ScanCur {
  const u8* base;  // record payload start
  i64 len;         // payload length
  i64 pos;         // parse position (monotonic, 0..len)
  i64 wpos;        // start of window w0 (multiple of 8)
  u64 w0, w1;      // windows [wpos, wpos+8) and [wpos+8, wpos+16)
  u32 crc;         // running ~crc state
  const uint32_t (*tab)[256];
};

TFR_HOSTDEV inline u64 cur_load(const ScanCur& c, i64 at) {
  if (at + 8 <= c.len) {
    u64 w;
    __builtin_memcpy(&w, c.base + at, 8);
    return w;
  }
  u64 w = 0;
  for (i64 b = at; b < c.len; ++b) w |= (u64)c.base[b] << (8 * (b - at));
  return w;
}

TFR_HOSTDEV inline void cur_init(ScanCur& c, const u8* base, i64 len,
                                 const uint32_t (*tab)[256]) {
  c.base = base;
  c.len = len;
  c.pos = 0;
  c.wpos = 0;
  c.tab = tab;
  c.crc = 0xFFFFFFFFu;
  c.w0 = cur_load(c, 0);
  c.w1 = cur_load(c, 8);
}

// Invariant: wpos <= pos <= wpos + 8 (so [pos, pos+8) is inside w0/w1).
TFR_HOSTDEV inline u64 cur_peek(const ScanCur& c) {
  int sh = (int)(c.pos - c.wpos) * 8;
  if (sh == 0) return c.w0;
  if (sh == 64) return c.w1;
  return (c.w0 >> sh) | (c.w1 << (64 - sh));
}

TFR_HOSTDEV inline void cur_advance(ScanCur& c, i64 k) {
  c.pos += k;
  while (c.pos > c.wpos + 8) {
    if (c.wpos + 8 <= c.len) c.crc = crc32c_step8(c.crc, c.w0, c.tab);
    c.w0 = c.w1;
    c.wpos += 8;
    c.w1 = cur_load(c, c.wpos + 8);
  }
}

// CRC of any bytes not yet folded (tail window + sub-8 remainder) -> final.
TFR_HOSTDEV inline u32 cur_finish_crc(ScanCur& c) {
  while (c.wpos + 8 <= c.len) {
    c.crc = crc32c_step8(c.crc, c.w0, c.tab);
    c.w0 = c.w1;
    c.wpos += 8;
    c.w1 = cur_load(c, c.wpos + 8);
  }
  u32 crc = c.crc;
  for (i64 b = c.wpos; b < c.len; ++b)
    crc = c.tab[0][(crc ^ c.base[b]) & 0xFF] ^ (crc >> 8);
  return ~crc;
}

TFR_HOSTDEV inline bool cur_varint(ScanCur& c, i64 end, u64* out) {
  i64 avail = end - c.pos;
  if (avail <= 0) return false;
  u64 w = cur_peek(c);
  u64 stops = ~w & kMsbMask;
  if (stops) {
    int nb = (ctz64(stops) >> 3) + 1;
    if (nb > avail) return false;  // terminator past the region end
    u64 v = 0;
    for (int i = 0; i < nb; ++i) v |= ((w >> (8 * i)) & 0x7F) << (7 * i);
    *out = v;
    cur_advance(c, nb);
    return true;
  }
  if (avail < 9) return false;
  u64 v = 0;
  for (int i = 0; i < 8; ++i) v |= ((w >> (8 * i)) & 0x7F) << (7 * i);
  cur_advance(c, 8);
  for (int shift = 56; shift < 64 && c.pos < end; shift += 7) {
    u8 b = (u8)cur_peek(c);
    v |= (u64)(b & 0x7F) << shift;
    cur_advance(c, 1);
    if (!(b & 0x80)) {
      *out = v;
      return true;
    }
  }
  return false;
}

TFR_HOSTDEV inline int32_t cur_skip_field(ScanCur& c, i64 end, u32 wt) {
  u64 tmp;
  switch (wt) {
    case 0:
      return cur_varint(c, end, &tmp) ? ERR_OK : ERR_BAD_VARINT;
    case 1:
      if (end - c.pos < 8) return ERR_TRUNCATED;
      cur_advance(c, 8);
      return ERR_OK;
    case 2:
      if (!cur_varint(c, end, &tmp)) return ERR_BAD_VARINT;
      if ((u64)(end - c.pos) < tmp) return ERR_TRUNCATED;
      cur_advance(c, (i64)tmp);
      return ERR_OK;
    case 5:
      if (end - c.pos < 4) return ERR_TRUNCATED;
      cur_advance(c, 4);
      return ERR_OK;
    default:
      return ERR_BAD_WIRETYPE;
  }
}

// Count varint terminators (MSB clear) in the next n bytes, window-wise.
TFR_HOSTDEV inline i64 cur_count_terms(ScanCur& c, i64 n) {
  i64 cnt = 0;
  while (n >= 8) {
    cnt += popcount64(~cur_peek(c) & kMsbMask);
    cur_advance(c, 8);
    n -= 8;
  }
  if (n > 0) {
    u64 mask = kMsbMask >> (8 * (8 - n));
    cnt += popcount64(~cur_peek(c) & mask);
    cur_advance(c, n);
  }
  return cnt;
}

TFR_HOSTDEV inline int32_t cur_count_list_body(ScanCur& c, i64 end, int32_t kind,
                                               i64* nvals, i64* nbytes) {
  while (c.pos < end) {
    u64 tag;
    if (!cur_varint(c, end, &tag)) return ERR_BAD_VARINT;
    u32 fieldno = (u32)(tag >> 3), wt = (u32)(tag & 7);
    if (fieldno != 1) {
      int32_t rc = cur_skip_field(c, end, wt);
      if (rc != ERR_OK) return rc;
      continue;
    }
    if (kind == KIND_BYTES) {
      if (wt != 2) return ERR_BAD_WIRETYPE;
      u64 blen;
      if (!cur_varint(c, end, &blen)) return ERR_BAD_VARINT;
      if ((u64)(end - c.pos) < blen) return ERR_TRUNCATED;
      *nvals += 1;
      *nbytes += (i64)blen;
      cur_advance(c, (i64)blen);
    } else if (kind == KIND_FLOAT) {
      if (wt == 2) {
        u64 blen;
        if (!cur_varint(c, end, &blen)) return ERR_BAD_VARINT;
        if ((u64)(end - c.pos) < blen) return ERR_TRUNCATED;
        *nvals += (i64)(blen / 4);
        cur_advance(c, (i64)blen);
      } else if (wt == 5) {
        if (end - c.pos < 4) return ERR_TRUNCATED;
        *nvals += 1;
        cur_advance(c, 4);
      } else {
        return ERR_BAD_WIRETYPE;
      }
    } else {  // KIND_INT64
      if (wt == 2) {
        u64 blen;
        if (!cur_varint(c, end, &blen)) return ERR_BAD_VARINT;
        if ((u64)(end - c.pos) < blen) return ERR_TRUNCATED;
        if (blen && (c.base[c.pos + (i64)blen - 1] & 0x80))
          return ERR_BAD_VARINT;  // packed run ends mid-varint
        *nvals += cur_count_terms(c, (i64)blen);
      } else if (wt == 0) {
        u64 v;
        if (!cur_varint(c, end, &v)) return ERR_BAD_VARINT;
        *nvals += 1;
      } else {
        return ERR_BAD_WIRETYPE;
      }
    }
  }
  return ERR_OK;
}

TFR_HOSTDEV inline int32_t cur_scan_feature_body(ScanCur& c, i64 end,
                                                 int32_t expect_kind,
                                                 int32_t* kind_found, i64* nvals,
                                                 i64* nbytes) {
  while (c.pos < end) {
    u64 tag;
    if (!cur_varint(c, end, &tag)) return ERR_BAD_VARINT;
    u32 fieldno = (u32)(tag >> 3), wt = (u32)(tag & 7);
    if (fieldno >= 1 && fieldno <= 3 && wt == 2) {
      u64 blen;
      if (!cur_varint(c, end, &blen)) return ERR_BAD_VARINT;
      if ((u64)(end - c.pos) < blen) return ERR_TRUNCATED;
      *kind_found = (int32_t)fieldno;
      if (expect_kind >= 0 && (int32_t)fieldno != expect_kind)
        return ERR_KIND_MISMATCH;
      int32_t rc = cur_count_list_body(c, c.pos + (i64)blen, (int32_t)fieldno,
                                       nvals, nbytes);
      if (rc != ERR_OK) return rc;
    } else {
      int32_t rc = cur_skip_field(c, end, wt);
      if (rc != ERR_OK) return rc;
    }
  }
  return ERR_OK;
}

TFR_HOSTDEV inline int32_t cur_scan_features_body(ScanCur& c, i64 end, i64 data_rel,
                                                  const SchemaView& schema,
                                                  int want_seq, FieldStat* stats) {
  i64 body_start = c.pos;
  while (c.pos < end) {
    u64 tag;
    if (!cur_varint(c, end, &tag)) return ERR_BAD_VARINT;
    u32 fieldno = (u32)(tag >> 3), wt = (u32)(tag & 7);
    if (fieldno != 1 || wt != 2) {
      int32_t rc = cur_skip_field(c, end, wt);
      if (rc != ERR_OK) return rc;
      continue;
    }
    u64 entry_len;
    if (!cur_varint(c, end, &entry_len)) return ERR_BAD_VARINT;
    if ((u64)(end - c.pos) < entry_len) return ERR_TRUNCATED;
    i64 ee = c.pos + (i64)entry_len;
    int f = -1;
    bool key_seen = false;
    bool val_seen = false;
    while (c.pos < ee) {
      u64 etag;
      if (!cur_varint(c, ee, &etag)) return ERR_BAD_VARINT;
      u32 efn = (u32)(etag >> 3), ewt = (u32)(etag & 7);
      if (efn == 1 && ewt == 2) {  // key
        // a key AFTER the value would override it (protobuf last-field-wins)
        // but the value was already consumed forward-only: redo two-pass
        if (val_seen) return ERR_RETRY_UNFUSED;
        u64 klen;
        if (!cur_varint(c, ee, &klen)) return ERR_BAD_VARINT;
        if ((u64)(ee - c.pos) < klen) return ERR_TRUNCATED;
        f = schema_find(schema, c.base + c.pos, (int)klen, want_seq);
        key_seen = true;
        cur_advance(c, (i64)klen);
      } else if (efn == 2 && ewt == 2) {  // value (Feature / FeatureList)
        if (!key_seen) return ERR_RETRY_UNFUSED;  // value before key: bail
        val_seen = true;
        u64 vlen;
        if (!cur_varint(c, ee, &vlen)) return ERR_BAD_VARINT;
        if ((u64)(ee - c.pos) < vlen) return ERR_TRUNCATED;
        i64 ve = c.pos + (i64)vlen;
        if (f < 0) {  // unknown feature: consume body (CRC still covers it)
          cur_advance(c, (i64)vlen);
          continue;
        }
        FieldStat* st = &stats[f];
        // duplicate map keys: last-entry-wins (see scan_features_body) —
        // reset the accumulators so only this body's counts survive
        st->nvals = 0;
        st->nbytes = 0;
        st->nlists = 0;
        st->kind_found = 0;
        st->err = 0;
        st->pos = data_rel + c.pos;
        st->len = (i64)vlen;
        if (!want_seq) {
          int32_t rc = cur_scan_feature_body(c, ve, schema.fields[f].kind,
                                             &st->kind_found, &st->nvals,
                                             &st->nbytes);
          if (rc == ERR_RETRY_UNFUSED) return rc;
          if (rc != ERR_OK) st->err = rc;  // mirror two-pass: record and go on
        } else {
          while (c.pos < ve && st->err == ERR_OK) {
            // FeatureList: repeated Feature (field 1)
            u64 ltag;
            if (!cur_varint(c, ve, &ltag)) {
              st->err = ERR_BAD_VARINT;
              break;
            }
            u32 lfn = (u32)(ltag >> 3), lwt = (u32)(ltag & 7);
            if (lfn == 1 && lwt == 2) {
              u64 flen;
              if (!cur_varint(c, ve, &flen) || (u64)(ve - c.pos) < flen) {
                st->err = ERR_TRUNCATED;
                break;
              }
              st->nlists += 1;
              int32_t kf = 0;
              int32_t rc = cur_scan_feature_body(c, c.pos + (i64)flen,
                                                 schema.fields[f].kind, &kf,
                                                 &st->nvals, &st->nbytes);
              if (rc == ERR_RETRY_UNFUSED) return rc;
              if (rc != ERR_OK) {
                st->err = rc;
                break;
              }
              if (kf) st->kind_found = kf;
            } else {
              int32_t rc = cur_skip_field(c, ve, lwt);
              if (rc != ERR_OK) {
                st->err = ERR_TRUNCATED;
                break;
              }
            }
          }
        }
        if (c.pos < ve) cur_advance(c, ve - c.pos);  // resync after stat error
      } else {
        int32_t rc = cur_skip_field(c, ee, ewt);
        if (rc != ERR_OK) return rc;
      }
    }
    if (f >= 0 && !val_seen) {
      // key with no value field: protobuf map semantics give the default
      // (empty) Feature — the feature IS present (two-pass form behavior:
      // scan_features_body sets pos = features-body start, len 0). Also a
      // last-wins reset: an earlier duplicate's counts must not survive.
      FieldStat* st = &stats[f];
      st->pos = data_rel + body_start;
      st->len = 0;
      st->nvals = 0;
      st->nbytes = 0;
      st->nlists = 0;
      st->kind_found = 0;
      st->err = 0;
    }
  }
  return ERR_OK;
}

// Fused pass-A entry: fills stats AND returns the payload CRC32C through
// *crc_out (valid only when the return code is ERR_OK). abs_off as in
// scan_record. ERR_RETRY_UNFUSED => caller must redo with the two-pass form.
TFR_HOSTDEV inline int32_t scan_record_fused(const u8* data, i64 abs_off,
                                             i64 rec_len, int32_t fmt,
                                             const SchemaView& schema,
                                             FieldStat* stats, u32* crc_out,
                                             const uint32_t (*tab)[256]) {
  ScanCur c;
  cur_init(c, data + abs_off, rec_len, tab);
  while (c.pos < rec_len) {
    u64 tag;
    if (!cur_varint(c, rec_len, &tag)) return ERR_BAD_VARINT;
    u32 fieldno = (u32)(tag >> 3), wt = (u32)(tag & 7);
    bool is_features = fieldno == 1;
    bool is_fl = (fmt == FMT_SEQUENCE && fieldno == 2);
    if ((is_features || is_fl) && wt == 2) {
      u64 blen;
      if (!cur_varint(c, rec_len, &blen)) return ERR_BAD_VARINT;
      if ((u64)(rec_len - c.pos) < blen) return ERR_TRUNCATED;
      int32_t rc = cur_scan_features_body(c, c.pos + (i64)blen, abs_off, schema,
                                          is_fl ? 1 : 0, stats);
      if (rc != ERR_OK) return rc;
    } else {
      int32_t rc = cur_skip_field(c, rec_len, wt);
      if (rc != ERR_OK) return rc;
    }
  }
  *crc_out = cur_finish_crc(c);
  return ERR_OK;
}

// ---------------------------------------------------------------------------
// Decode pass B: value extraction for one (record, field), given the body
// extent from pass A and destination offsets from the prefix sums.
// ---------------------------------------------------------------------------

struct DecodeDst {
  i64* i64_vals;
  float* f32_vals;
  u8* bytes_data;   // flat string bytes
  i64* elem_len;    // per-string byte length, later prefix-summed in Python
  i64* sub_count;   // per-sub-list value count (seq), later prefix-summed
};

// Extracts one list body's values. Cursors are advanced versions of the
// pass-A counters: *vi = next value slot, *bi = next byte offset.
TFR_HOSTDEV inline int32_t extract_list_body(const u8* p, const u8* end, int32_t kind,
                                             const DecodeDst& dst, i64* vi, i64* bi) {
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) return ERR_BAD_VARINT;
    u32 fieldno = static_cast<u32>(tag >> 3);
    u32 wt = static_cast<u32>(tag & 7);
    if (fieldno != 1) {
      p = skip_field(p, end, wt);
      if (!p) return ERR_TRUNCATED;
      continue;
    }
    if (kind == KIND_BYTES) {
      u64 len;
      p = read_varint(p, end, &len);
      if (!p || static_cast<u64>(end - p) < len) return ERR_TRUNCATED;
      // word-wise copy (unaligned u64 loads/stores are fine on gfx950 and
      // x86; a byte loop costs 8x the memory instructions)
      u8* d = dst.bytes_data + *bi;
      u64 i = 0;
      for (; i + 8 <= len; i += 8) {
        u64 w;
        __builtin_memcpy(&w, p + i, 8);
        __builtin_memcpy(d + i, &w, 8);
      }
      for (; i < len; ++i) d[i] = p[i];
      dst.elem_len[*vi] = static_cast<i64>(len);
      *bi += static_cast<i64>(len);
      *vi += 1;
      p += len;
    } else if (kind == KIND_FLOAT) {
      if (wt == 2) {
        u64 len;
        p = read_varint(p, end, &len);
        if (!p || static_cast<u64>(end - p) < len) return ERR_TRUNCATED;
        u64 n = len / 4;
        __builtin_memcpy(dst.f32_vals + *vi, p, n * 4);  // packed LE floats
        *vi += static_cast<i64>(n);
        p += len;
      } else {  // fixed32
        if (end - p < 4) return ERR_TRUNCATED;
        float v;
        __builtin_memcpy(&v, p, 4);
        dst.f32_vals[(*vi)++] = v;
        p += 4;
      }
    } else {  // INT64
      if (wt == 2) {
        u64 len;
        p = read_varint(p, end, &len);
        if (!p || static_cast<u64>(end - p) < len) return ERR_TRUNCATED;
        const u8* q = p;
        const u8* qe = p + len;
        while (q < qe) {
          u64 v;
          q = read_varint(q, qe, &v);
          if (!q) return ERR_BAD_VARINT;
          dst.i64_vals[(*vi)++] = static_cast<i64>(v);
        }
        p = qe;
      } else {
        u64 v;
        p = read_varint(p, end, &v);
        if (!p) return ERR_BAD_VARINT;
        dst.i64_vals[(*vi)++] = static_cast<i64>(v);
      }
    }
  }
  return ERR_OK;
}

TFR_HOSTDEV inline int32_t extract_feature_body(const u8* p, const u8* end,
                                                int32_t kind, const DecodeDst& dst,
                                                i64* vi, i64* bi) {
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) return ERR_BAD_VARINT;
    u32 fieldno = static_cast<u32>(tag >> 3);
    u32 wt = static_cast<u32>(tag & 7);
    if (fieldno >= 1 && fieldno <= 3 && wt == 2) {
      u64 len;
      p = read_varint(p, end, &len);
      if (!p || static_cast<u64>(end - p) < len) return ERR_TRUNCATED;
      int32_t rc = extract_list_body(p, p + len, kind, dst, vi, bi);
      if (rc != ERR_OK) return rc;
      p += len;
    } else {
      p = skip_field(p, end, wt);
      if (!p) return ERR_TRUNCATED;
    }
  }
  return ERR_OK;
}

// Pass B for one (record, field): body extent [pos, pos+len) from pass A.
// val_base / byte_base / list_base are this row's starting slots from the
// prefix sums. For seq fields, also emits per-sub-list counts.
TFR_HOSTDEV inline int32_t extract_field(const u8* data, i64 pos, i64 len,
                                         int32_t kind, int32_t is_seq,
                                         const DecodeDst& dst, i64 val_base,
                                         i64 byte_base, i64 list_base) {
  if (pos < 0) return ERR_OK;
  const u8* p = data + pos;
  const u8* end = p + len;
  i64 vi = val_base;
  i64 bi = byte_base;
  if (!is_seq) return extract_feature_body(p, end, kind, dst, &vi, &bi);
  // FeatureList: repeated Feature
  i64 li = list_base;
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) return ERR_BAD_VARINT;
    u32 fieldno = static_cast<u32>(tag >> 3);
    u32 wt = static_cast<u32>(tag & 7);
    if (fieldno == 1 && wt == 2) {
      u64 flen;
      p = read_varint(p, end, &flen);
      if (!p || static_cast<u64>(end - p) < flen) return ERR_TRUNCATED;
      i64 v_before = vi;
      int32_t rc = extract_feature_body(p, p + flen, kind, dst, &vi, &bi);
      if (rc != ERR_OK) return rc;
      dst.sub_count[li++] = vi - v_before;
      p += flen;
    } else {
      p = skip_field(p, end, wt);
      if (!p) return ERR_TRUNCATED;
    }
  }
  return ERR_OK;
}

// ---------------------------------------------------------------------------
// Encode: columnar wire-form -> serialized Example/SequenceExample payloads.
// Two passes (size, then emit) so the GPU can prefix-sum record offsets in
// between; the per-record logic is identical for both and shared with host.
// Mirrors the reference serializer's feature construction
// (TFRecordSerializer.scala:20-60, builders :182-207).
// ---------------------------------------------------------------------------

struct FieldColumn {
  int32_t kind;
  int32_t is_seq;
  const u8* presence;    // u8[R]; 0 => feature omitted (nullable null)
  const i64* row_off;    // i64[R+1] cumulative VALUES per row
  const i64* list_off;   // i64[R+1] cumulative sub-lists per row (seq only)
  const i64* sub_off;    // i64[L+1] cumulative values per sub-list (seq only)
  const i64* elem_off;   // i64[E+1] cumulative bytes per string (bytes only)
  const u8* bytes_data;
  const i64* i64_vals;
  const float* f32_vals;
};

// Size of one list payload (the BytesList/FloatList/Int64List *body*), for
// values [v0, v1) of a column.
TFR_HOSTDEV inline i64 list_body_size(const FieldColumn& c, i64 v0, i64 v1) {
  if (c.kind == KIND_FLOAT) {
    i64 n = v1 - v0;
    return n ? (1 + varint_size(static_cast<u64>(4 * n)) + 4 * n) : 0;
  }
  if (c.kind == KIND_INT64) {
    i64 packed = 0;
    for (i64 v = v0; v < v1; ++v)
      packed += varint_size(static_cast<u64>(c.i64_vals[v]));
    return packed ? (1 + varint_size(static_cast<u64>(packed)) + packed) : 0;
  }
  // bytes: repeated strings, each tag+len+data
  i64 sz = 0;
  for (i64 v = v0; v < v1; ++v) {
    i64 blen = c.elem_off[v + 1] - c.elem_off[v];
    sz += 1 + varint_size(static_cast<u64>(blen)) + blen;
  }
  return sz;
}

// Feature message body size for values [v0, v1): kind tag + len + list body.
TFR_HOSTDEV inline i64 feature_body_size(const FieldColumn& c, i64 v0, i64 v1) {
  i64 body = list_body_size(c, v0, v1);
  return 1 + varint_size(static_cast<u64>(body)) + body;
}

TFR_HOSTDEV inline u8* emit_list_body(u8* o, const FieldColumn& c, i64 v0, i64 v1) {
  if (c.kind == KIND_FLOAT) {
    i64 n = v1 - v0;
    if (n) {
      *o++ = 0x0A;  // field 1, wiretype 2 (packed)
      o = write_varint(o, static_cast<u64>(4 * n));
      __builtin_memcpy(o, c.f32_vals + v0, static_cast<size_t>(4 * n));
      o += 4 * n;
    }
  } else if (c.kind == KIND_INT64) {
    i64 packed = 0;
    for (i64 v = v0; v < v1; ++v)
      packed += varint_size(static_cast<u64>(c.i64_vals[v]));
    if (packed) {
      *o++ = 0x0A;
      o = write_varint(o, static_cast<u64>(packed));
      for (i64 v = v0; v < v1; ++v)
        o = write_varint(o, static_cast<u64>(c.i64_vals[v]));
    }
  } else {
    for (i64 v = v0; v < v1; ++v) {
      i64 b0 = c.elem_off[v];
      i64 blen = c.elem_off[v + 1] - b0;
      *o++ = 0x0A;
      o = write_varint(o, static_cast<u64>(blen));
      for (i64 i = 0; i < blen; ++i) o[i] = c.bytes_data[b0 + i];
      o += blen;
    }
  }
  return o;
}

TFR_HOSTDEV inline u8* emit_feature_body(u8* o, const FieldColumn& c, i64 v0, i64 v1) {
  i64 body = list_body_size(c, v0, v1);
  *o++ = static_cast<u8>((c.kind << 3) | 2);  // bytes_list=0x0A float=0x12 int64=0x1A
  o = write_varint(o, static_cast<u64>(body));
  return emit_list_body(o, c, v0, v1);
}

// map-entry body (key + Feature value) size for a non-seq field of row r.
TFR_HOSTDEV inline i64 features_entry_size(const FieldColumn& c, const SchemaView& s,
                                           int f, i64 v0, i64 v1) {
  i64 fb = feature_body_size(c, v0, v1);
  i64 klen = s.name_len(f);
  return (1 + varint_size(static_cast<u64>(klen)) + klen) +
         (1 + varint_size(static_cast<u64>(fb)) + fb);
}

TFR_HOSTDEV inline u8* emit_features_entry(u8* o, const FieldColumn& c,
                                           const SchemaView& s, int f, i64 v0, i64 v1) {
  i64 fb = feature_body_size(c, v0, v1);
  i64 klen = s.name_len(f);
  *o++ = 0x0A;  // key
  o = write_varint(o, static_cast<u64>(klen));
  const u8* nm = s.name(f);
  for (i64 i = 0; i < klen; ++i) o[i] = nm[i];
  o += klen;
  *o++ = 0x12;  // value (Feature)
  o = write_varint(o, static_cast<u64>(fb));
  return emit_feature_body(o, c, v0, v1);
}

// FeatureList value body (repeated Feature) for row r of a seq field.
TFR_HOSTDEV inline i64 feature_list_body_size(const FieldColumn& c, i64 r) {
  i64 sz = 0;
  for (i64 j = c.list_off[r]; j < c.list_off[r + 1]; ++j) {
    i64 fb = feature_body_size(c, c.sub_off[j], c.sub_off[j + 1]);
    sz += 1 + varint_size(static_cast<u64>(fb)) + fb;
  }
  return sz;
}

TFR_HOSTDEV inline u8* emit_feature_list_body(u8* o, const FieldColumn& c, i64 r) {
  for (i64 j = c.list_off[r]; j < c.list_off[r + 1]; ++j) {
    i64 fb = feature_body_size(c, c.sub_off[j], c.sub_off[j + 1]);
    *o++ = 0x0A;  // FeatureList.feature
    o = write_varint(o, static_cast<u64>(fb));
    o = emit_feature_body(o, c, c.sub_off[j], c.sub_off[j + 1]);
  }
  return o;
}

// Seq map-entry (key + FeatureList) size for row r.
TFR_HOSTDEV inline i64 feature_lists_entry_size(const FieldColumn& c,
                                                const SchemaView& s, int f, i64 r) {
  i64 flb = feature_list_body_size(c, r);
  i64 klen = s.name_len(f);
  return (1 + varint_size(static_cast<u64>(klen)) + klen) +
         (1 + varint_size(static_cast<u64>(flb)) + flb);
}

TFR_HOSTDEV inline u8* emit_feature_lists_entry(u8* o, const FieldColumn& c,
                                                const SchemaView& s, int f, i64 r) {
  i64 flb = feature_list_body_size(c, r);
  i64 klen = s.name_len(f);
  *o++ = 0x0A;
  o = write_varint(o, static_cast<u64>(klen));
  const u8* nm = s.name(f);
  for (i64 i = 0; i < klen; ++i) o[i] = nm[i];
  o += klen;
  *o++ = 0x12;
  o = write_varint(o, static_cast<u64>(flb));
  return emit_feature_list_body(o, c, r);
}

// Full record payload size for row r (Example or SequenceExample).
TFR_HOSTDEV inline i64 record_payload_size(const FieldColumn* cols,
                                           const SchemaView& s, int32_t fmt, i64 r) {
  i64 ctx_body = 0;  // Features body (Example.features / SequenceExample.context)
  i64 fl_body = 0;   // FeatureLists body
  for (int f = 0; f < s.nfields; ++f) {
    const FieldColumn& c = cols[f];
    if (!c.presence[r]) continue;
    if (!c.is_seq) {
      i64 e = features_entry_size(c, s, f, c.row_off[r], c.row_off[r + 1]);
      ctx_body += 1 + varint_size(static_cast<u64>(e)) + e;
    } else {
      i64 e = feature_lists_entry_size(c, s, f, r);
      fl_body += 1 + varint_size(static_cast<u64>(e)) + e;
    }
  }
  i64 total = 0;
  if (fmt == FMT_EXAMPLE) {
    // Always emit the features submessage (tag+len), even when empty — matches
    // a builder with features always set.
    total += 1 + varint_size(static_cast<u64>(ctx_body)) + ctx_body;
  } else {
    if (ctx_body) total += 1 + varint_size(static_cast<u64>(ctx_body)) + ctx_body;
    total += 1 + varint_size(static_cast<u64>(fl_body)) + fl_body;
  }
  return total;
}

TFR_HOSTDEV inline u8* emit_record_payload(u8* o, const FieldColumn* cols,
                                           const SchemaView& s, int32_t fmt, i64 r) {
  i64 ctx_body = 0, fl_body = 0;
  for (int f = 0; f < s.nfields; ++f) {
    const FieldColumn& c = cols[f];
    if (!c.presence[r]) continue;
    if (!c.is_seq) {
      i64 e = features_entry_size(c, s, f, c.row_off[r], c.row_off[r + 1]);
      ctx_body += 1 + varint_size(static_cast<u64>(e)) + e;
    } else {
      i64 e = feature_lists_entry_size(c, s, f, r);
      fl_body += 1 + varint_size(static_cast<u64>(e)) + e;
    }
  }
  auto emit_ctx = [&](u8* oo) {
    for (int f = 0; f < s.nfields; ++f) {
      const FieldColumn& c = cols[f];
      if (!c.presence[r] || c.is_seq) continue;
      i64 e = features_entry_size(c, s, f, c.row_off[r], c.row_off[r + 1]);
      *oo++ = 0x0A;  // Features.feature map entry
      oo = write_varint(oo, static_cast<u64>(e));
      oo = emit_features_entry(oo, c, s, f, c.row_off[r], c.row_off[r + 1]);
    }
    return oo;
  };
  auto emit_fl = [&](u8* oo) {
    for (int f = 0; f < s.nfields; ++f) {
      const FieldColumn& c = cols[f];
      if (!c.presence[r] || !c.is_seq) continue;
      i64 e = feature_lists_entry_size(c, s, f, r);
      *oo++ = 0x0A;  // FeatureLists.feature_list map entry
      oo = write_varint(oo, static_cast<u64>(e));
      oo = emit_feature_lists_entry(oo, c, s, f, r);
    }
    return oo;
  };
  if (fmt == FMT_EXAMPLE) {
    *o++ = 0x0A;  // Example.features
    o = write_varint(o, static_cast<u64>(ctx_body));
    o = emit_ctx(o);
  } else {
    if (ctx_body) {
      *o++ = 0x0A;  // SequenceExample.context
      o = write_varint(o, static_cast<u64>(ctx_body));
      o = emit_ctx(o);
    }
    *o++ = 0x12;  // SequenceExample.feature_lists
    o = write_varint(o, static_cast<u64>(fl_body));
    o = emit_fl(o);
  }
  return o;
}

// ---------------------------------------------------------------------------
// Frame emit: payload -> [u64 len][masked crc(len)][payload][masked crc(payload)]
// at an absolute offset. Payload must already be in place at frame_off + 12.
// ---------------------------------------------------------------------------

// ---------------------------------------------------------------------------
// Fused emit + CRC (write cursor): the mirror image of ScanCur. Output bytes
// accumulate in an 8-byte register window; each full window is stored ONCE
// (one unaligned u64 store instead of eight byte stores) and folded into the
// running payload CRC32C — the separate frame-CRC step used to re-read the
// whole emitted payload. Byte-exactness vs the plain emitters is guaranteed
// by construction (same byte sequence, different batching) and covered by
// the golden-vector / protobuf-interop / GPU-vs-host equality tests.
// ---------------------------------------------------------------------------

struct WriteCur {
  u8* base;   // payload destination
  i64 pos;    // bytes emitted
  i64 wpos;   // bytes stored + CRC'd (multiple of 8)
  u64 win;    // pending low (pos - wpos) bytes
  u32 crc;    // running ~crc
  const uint32_t (*tab)[256];
};

TFR_HOSTDEV inline void wcur_init(WriteCur& w, u8* base,
                                  const uint32_t (*tab)[256]) {
  w.base = base;
  w.pos = 0;
  w.wpos = 0;
  w.win = 0;
  w.crc = 0xFFFFFFFFu;
  w.tab = tab;
}

TFR_HOSTDEV inline void wcur_flush8(WriteCur& w) {
  __builtin_memcpy(w.base + w.wpos, &w.win, 8);  // little-endian everywhere
  w.crc = crc32c_step8(w.crc, w.win, w.tab);
  w.wpos += 8;
  w.win = 0;
}

TFR_HOSTDEV inline void wcur_put(WriteCur& w, u8 b) {
  w.win |= (u64)b << (8 * (w.pos - w.wpos));
  if (++w.pos - w.wpos == 8) wcur_flush8(w);
}

TFR_HOSTDEV inline void wcur_varint(WriteCur& w, u64 v) {
  while (v >= 0x80) {
    wcur_put(w, (u8)(v | 0x80));
    v >>= 7;
  }
  wcur_put(w, (u8)v);
}

TFR_HOSTDEV inline void wcur_bytes(WriteCur& w, const u8* src, i64 n) {
  while (n > 0 && (w.pos - w.wpos) != 0) {  // fill to the window boundary
    wcur_put(w, *src++);
    --n;
  }
  while (n >= 8) {  // window empty: stream whole words through it
    __builtin_memcpy(&w.win, src, 8);
    w.pos += 8;
    wcur_flush8(w);
    src += 8;
    n -= 8;
  }
  while (n-- > 0) wcur_put(w, *src++);
}

// Stores the partial tail window and returns the final payload CRC32C.
TFR_HOSTDEV inline u32 wcur_finish(WriteCur& w) {
  int rem = (int)(w.pos - w.wpos);
  if (!rem) return ~w.crc;
  __builtin_memcpy(w.base + w.wpos, &w.win, (size_t)rem);
  u32 crc = w.crc;
  for (int i = 0; i < rem; ++i)
    crc = w.tab[0][(crc ^ (u8)(w.win >> (8 * i))) & 0xFF] ^ (crc >> 8);
  return ~crc;
}

TFR_HOSTDEV inline void wcur_list_body(WriteCur& w, const FieldColumn& c,
                                       i64 v0, i64 v1) {
  if (c.kind == KIND_FLOAT) {
    i64 n = v1 - v0;
    if (n) {
      wcur_put(w, 0x0A);  // field 1, wiretype 2 (packed)
      wcur_varint(w, (u64)(4 * n));
      wcur_bytes(w, (const u8*)(c.f32_vals + v0), 4 * n);
    }
  } else if (c.kind == KIND_INT64) {
    i64 packed = 0;
    for (i64 v = v0; v < v1; ++v)
      packed += varint_size((u64)c.i64_vals[v]);
    if (packed) {
      wcur_put(w, 0x0A);
      wcur_varint(w, (u64)packed);
      for (i64 v = v0; v < v1; ++v) wcur_varint(w, (u64)c.i64_vals[v]);
    }
  } else {
    for (i64 v = v0; v < v1; ++v) {
      i64 b0 = c.elem_off[v];
      i64 blen = c.elem_off[v + 1] - b0;
      wcur_put(w, 0x0A);
      wcur_varint(w, (u64)blen);
      wcur_bytes(w, c.bytes_data + b0, blen);
    }
  }
}

TFR_HOSTDEV inline void wcur_feature_body(WriteCur& w, const FieldColumn& c,
                                          i64 v0, i64 v1) {
  i64 body = list_body_size(c, v0, v1);
  wcur_put(w, (u8)((c.kind << 3) | 2));
  wcur_varint(w, (u64)body);
  wcur_list_body(w, c, v0, v1);
}

TFR_HOSTDEV inline void wcur_features_entry(WriteCur& w, const FieldColumn& c,
                                            const SchemaView& s, int f, i64 v0,
                                            i64 v1) {
  i64 fb = feature_body_size(c, v0, v1);
  i64 klen = s.name_len(f);
  wcur_put(w, 0x0A);  // key
  wcur_varint(w, (u64)klen);
  wcur_bytes(w, s.name(f), klen);
  wcur_put(w, 0x12);  // value (Feature)
  wcur_varint(w, (u64)fb);
  wcur_feature_body(w, c, v0, v1);
}

TFR_HOSTDEV inline void wcur_feature_list_body(WriteCur& w,
                                               const FieldColumn& c, i64 r) {
  for (i64 j = c.list_off[r]; j < c.list_off[r + 1]; ++j) {
    i64 fb = feature_body_size(c, c.sub_off[j], c.sub_off[j + 1]);
    wcur_put(w, 0x0A);  // FeatureList.feature
    wcur_varint(w, (u64)fb);
    wcur_feature_body(w, c, c.sub_off[j], c.sub_off[j + 1]);
  }
}

TFR_HOSTDEV inline void wcur_feature_lists_entry(WriteCur& w,
                                                 const FieldColumn& c,
                                                 const SchemaView& s, int f,
                                                 i64 r) {
  i64 flb = feature_list_body_size(c, r);
  i64 klen = s.name_len(f);
  wcur_put(w, 0x0A);
  wcur_varint(w, (u64)klen);
  wcur_bytes(w, s.name(f), klen);
  wcur_put(w, 0x12);
  wcur_varint(w, (u64)flb);
  wcur_feature_list_body(w, c, r);
}

// Emits one record payload at `o`, returning its length; *crc_out gets the
// payload CRC32C accumulated from the same register windows the stores used.
TFR_HOSTDEV inline i64 emit_record_payload_fused(u8* o, const FieldColumn* cols,
                                                 const SchemaView& s, int32_t fmt,
                                                 i64 r, u32* crc_out,
                                                 const uint32_t (*tab)[256]) {
  i64 ctx_body = 0, fl_body = 0;
  for (int f = 0; f < s.nfields; ++f) {
    const FieldColumn& c = cols[f];
    if (!c.presence[r]) continue;
    if (!c.is_seq) {
      i64 e = features_entry_size(c, s, f, c.row_off[r], c.row_off[r + 1]);
      ctx_body += 1 + varint_size((u64)e) + e;
    } else {
      i64 e = feature_lists_entry_size(c, s, f, r);
      fl_body += 1 + varint_size((u64)e) + e;
    }
  }
  WriteCur w;
  wcur_init(w, o, tab);
  auto emit_ctx = [&]() {
    for (int f = 0; f < s.nfields; ++f) {
      const FieldColumn& c = cols[f];
      if (!c.presence[r] || c.is_seq) continue;
      i64 e = features_entry_size(c, s, f, c.row_off[r], c.row_off[r + 1]);
      wcur_put(w, 0x0A);  // Features.feature map entry
      wcur_varint(w, (u64)e);
      wcur_features_entry(w, c, s, f, c.row_off[r], c.row_off[r + 1]);
    }
  };
  auto emit_fl = [&]() {
    for (int f = 0; f < s.nfields; ++f) {
      const FieldColumn& c = cols[f];
      if (!c.presence[r] || !c.is_seq) continue;
      i64 e = feature_lists_entry_size(c, s, f, r);
      wcur_put(w, 0x0A);  // FeatureLists.feature_list map entry
      wcur_varint(w, (u64)e);
      wcur_feature_lists_entry(w, c, s, f, r);
    }
  };
  if (fmt == FMT_EXAMPLE) {
    wcur_put(w, 0x0A);  // Example.features
    wcur_varint(w, (u64)ctx_body);
    emit_ctx();
  } else {
    if (ctx_body) {
      wcur_put(w, 0x0A);  // SequenceExample.context
      wcur_varint(w, (u64)ctx_body);
      emit_ctx();
    }
    wcur_put(w, 0x12);  // SequenceExample.feature_lists
    wcur_varint(w, (u64)fl_body);
    emit_fl();
  }
  *crc_out = wcur_finish(w);
  return w.pos;
}

constexpr i64 kFrameOverhead = 16;  // 8 len + 4 crc + 4 crc

TFR_HOSTDEV inline void write_frame_header_footer(u8* file, i64 frame_off,
                                                  i64 payload_len,
                                                  const u32 (*tab)[256]) {
  u8* h = file + frame_off;
  u64 len_le = static_cast<u64>(payload_len);
  __builtin_memcpy(h, &len_le, 8);  // little-endian host & device
  u32 lc = mask_crc(crc32c_sw(h, 8, 0, tab));
  __builtin_memcpy(h + 8, &lc, 4);
  u32 dc = mask_crc(crc32c_sw(h + 12, static_cast<size_t>(payload_len), 0, tab));
  __builtin_memcpy(h + 12 + payload_len, &dc, 4);
}

// Variant for the fused emitter: the payload CRC is already known, so only
// the 8 header bytes are (re)read for their CRC — no payload re-read.
TFR_HOSTDEV inline void write_frame_header_footer_crc(u8* file, i64 frame_off,
                                                      i64 payload_len,
                                                      u32 payload_crc,
                                                      const u32 (*tab)[256]) {
  u8* h = file + frame_off;
  u64 len_le = static_cast<u64>(payload_len);
  __builtin_memcpy(h, &len_le, 8);
  u32 lc = mask_crc(crc32c_sw(h, 8, 0, tab));
  __builtin_memcpy(h + 8, &lc, 4);
  u32 dc = mask_crc(payload_crc);
  __builtin_memcpy(h + 12 + payload_len, &dc, 4);
}

}  // namespace tfrec
