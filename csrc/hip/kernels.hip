// gfx950 (CDNA4) kernels for the TFRecord codec + pybind11 launch wrappers.
//
// Design (SURVEY.md §2b native inventory, §7 step 3; docs/KERNELS.md):
//  - Records are the parallel axis: one record per lane, grid-stride loops
//    sized so a full file saturates 256 CUs (wave = 64 lanes; blocks of 256).
//  - Frame boundaries are discovered ON DEVICE: every byte position is
//    CRC-tested as a candidate head by a two-pass (count/emit) kernel whose
//    output is position-sorted by construction.
//  - The structure scan and the emit both FUSE the payload CRC32C into the
//    same register windows their loads/stores use (codec_core.h ScanCur /
//    WriteCur); CRC32C slicing-by-8 tables (8 KiB) are staged in LDS.
//  - Prefix sums between passes are rocprim decoupled-lookback scans with
//    strided input iterators (Python orchestration in engine/gpu.py).
//
// All parsing/emit logic is shared with the host via csrc/codec_core.h — the
// kernels only add the parallel decomposition and memory staging.

#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>
#include <rocprim/rocprim.hpp>

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstdint>
#include <map>
#include <mutex>
#include <stdexcept>
#include <string>
#include <vector>

#include "../codec_core.h"

namespace py = pybind11;
using namespace tfrec;

#define HIP_CHECK(expr)                                                          \
  do {                                                                           \
    hipError_t _e = (expr);                                                      \
    if (_e != hipSuccess)                                                        \
      throw std::runtime_error(std::string("HIP error: ") +                      \
                               hipGetErrorString(_e) + " at " __FILE__ ":" +     \
                               std::to_string(__LINE__));                        \
  } while (0)

namespace {

constexpr int kBlock = 256;

__device__ inline void stage_crc_tables(uint32_t (*lds)[256]) {
  const uint32_t* src = &kCrcTables.t[0][0];
  uint32_t* dst = &lds[0][0];
  for (int i = threadIdx.x; i < 8 * 256; i += blockDim.x) dst[i] = src[i];
  __syncthreads();
}

// Error buffers are int32[2]: [0] = codec error code, [1] = 1 + a record
// index that hit it (atomicMax keeps one deterministically; Python adds any
// launch base offset and includes the index in the raised message).
__device__ inline void set_err(int32_t* err, int32_t code, i64 r) {
  err[0] = code;
  i64 v = r + 1;
  atomicMax(&err[1], v > 0x7FFFFFFF ? (int32_t)0x7FFFFFFF : (int32_t)v);
}

// ---------------------------------------------------------------------------
// CRC verification: one record per lane. err_out[0] stays 0 on success, else
// holds 1 + first-bad-record index (atomicMin keeps the earliest).
// ---------------------------------------------------------------------------

__global__ void crc_verify_kernel(const u8* __restrict__ data,
                                  const i64* __restrict__ off,
                                  const i64* __restrict__ len, i64 R,
                                  unsigned long long* err_out) {
  __shared__ uint32_t tab[8][256];
  stage_crc_tables(tab);
  for (i64 r = blockIdx.x * (i64)blockDim.x + threadIdx.x; r < R;
       r += (i64)gridDim.x * blockDim.x) {
    const u8* h = data + off[r] - 12;  // frame header precedes payload
    u32 len_crc, data_crc;
    __builtin_memcpy(&len_crc, h + 8, 4);
    __builtin_memcpy(&data_crc, h + 12 + len[r], 4);
    bool ok = mask_crc(crc32c_sw(h, 8, 0, tab)) == len_crc &&
              mask_crc(crc32c_sw(h + 12, (size_t)len[r], 0, tab)) == data_crc;
    if (!ok) atomicMin(err_out, (unsigned long long)(r + 1));
  }
}

// ---------------------------------------------------------------------------
// Decode pass A: structure scan. Stats land directly in a global [R][F]
// FieldStat buffer (48 B = 6 int64 words each) — no per-thread scratch array;
// Python views the buffer as an int64 [R,F,6] tensor for torch prefix sums.
// ---------------------------------------------------------------------------

static_assert(sizeof(FieldStat) == 48, "FieldStat layout shared with Python");

// When crc_err != nullptr the frame CRCs (header + payload) are verified in
// the same pass, while the record bytes are L2-hot from the parse — a
// separate verify kernel costs an extra full-file HBM read and an extra
// host sync. crc_err[0] stays ~0ULL on success, else 1 + first bad record.
__global__ void scan_records_kernel(const u8* __restrict__ data,
                                    const i64* __restrict__ off,
                                    const i64* __restrict__ len, i64 R, int32_t fmt,
                                    const u8* __restrict__ schema_blob, int F,
                                    FieldStat* __restrict__ stats,
                                    int32_t* __restrict__ err,
                                    unsigned long long* crc_err) {
  __shared__ uint32_t tab[8][256];
  if (crc_err) stage_crc_tables(tab);
  SchemaView schema = schema_view(schema_blob);
  for (i64 r = blockIdx.x * (i64)blockDim.x + threadIdx.x; r < R;
       r += (i64)gridDim.x * blockDim.x) {
    FieldStat* st = stats + r * F;
    for (int f = 0; f < F; ++f) field_stat_clear(&st[f]);
    int32_t rc;
    if (crc_err) {
      // fused cursor pass: each payload window is loaded ONCE for both the
      // structure scan and the CRC (codec_core.h ScanCur)
      u32 payload_crc = 0;
      rc = scan_record_fused(data, off[r], len[r], fmt, schema, st,
                             &payload_crc, tab);
      if (rc == ERR_RETRY_UNFUSED) {  // value-before-key map entry: redo
        for (int f = 0; f < F; ++f) field_stat_clear(&st[f]);
        rc = scan_record(data, off[r], len[r], fmt, schema, st);
        payload_crc = crc32c_sw(data + off[r], (size_t)len[r], 0, tab);
      }
      if (rc != ERR_OK)  // parse bailed mid-record: recompute CRC plainly
        payload_crc = crc32c_sw(data + off[r], (size_t)len[r], 0, tab);
      const u8* h = data + off[r] - 12;
      u32 len_crc, data_crc;
      __builtin_memcpy(&len_crc, h + 8, 4);
      __builtin_memcpy(&data_crc, h + 12 + len[r], 4);
      bool ok = mask_crc(crc32c_sw(h, 8, 0, tab)) == len_crc &&
                mask_crc(payload_crc) == data_crc;
      if (!ok) atomicMin(crc_err, (unsigned long long)(r + 1));
    } else {
      rc = scan_record(data, off[r], len[r], fmt, schema, st);
    }
    if (rc != ERR_OK) set_err(err, rc, r);
    for (int f = 0; f < F; ++f)
      if (st[f].err != ERR_OK) set_err(err, st[f].err, r);
  }
}

// ---------------------------------------------------------------------------
// Decode pass B: value extraction, one (record, field) pair per lane.
// Row bases come from the exclusive prefix sums ([F][R+1], row r's base at
// index r, i.e. cumsum shifted by one).
// ---------------------------------------------------------------------------

struct DevFieldDst {
  int32_t kind;
  int32_t is_seq;
  i64* i64_vals;
  float* f32_vals;
  u8* bytes_data;
  i64* elem_len;
  i64* sub_count;
  const i64* val_base;   // [R+1] exclusive scan of nvals
  const i64* byte_base;  // [R+1] exclusive scan of nbytes
  const i64* list_base;  // [R+1] exclusive scan of nlists
};

__global__ void extract_fields_kernel(const u8* __restrict__ data, i64 R, int F,
                                      const FieldStat* __restrict__ stats,
                                      const DevFieldDst* __restrict__ metas,
                                      int32_t* __restrict__ err) {
  i64 total = R * F;
  for (i64 idx = blockIdx.x * (i64)blockDim.x + threadIdx.x; idx < total;
       idx += (i64)gridDim.x * blockDim.x) {
    i64 r = idx / F;
    int f = (int)(idx - r * F);
    const DevFieldDst& m = metas[f];
    const FieldStat& st = stats[r * F + f];
    DecodeDst dst{m.i64_vals, m.f32_vals, m.bytes_data, m.elem_len, m.sub_count};
    int32_t rc = extract_field(data, st.pos, st.len, m.kind, m.is_seq, dst,
                               m.val_base ? m.val_base[r] : 0,
                               m.byte_base ? m.byte_base[r] : 0,
                               m.list_base ? m.list_base[r] : 0);
    if (rc != ERR_OK) set_err(err, rc, r);
  }
}

// ---------------------------------------------------------------------------
// Encode: size pass (thread per record), emit pass, frame+CRC pass.
// ---------------------------------------------------------------------------

// The device column table is a variable-length FieldColumn array sized by
// the schema (no fixed field cap; schema.nfields drives every loop).

__global__ void size_records_kernel(const FieldColumn* __restrict__ cols,
                                    const u8* __restrict__ schema_blob, int32_t fmt,
                                    i64 R, i64* __restrict__ psize) {
  SchemaView schema = schema_view(schema_blob);
  for (i64 r = blockIdx.x * (i64)blockDim.x + threadIdx.x; r < R;
       r += (i64)gridDim.x * blockDim.x) {
    psize[r] = record_payload_size(cols, schema, fmt, r) + kFrameOverhead;
  }
}

// frame_off is the exclusive scan of psize (includes frame overhead).
// The frame header/footer CRC is computed HERE, right after the payload is
// emitted, while the bytes are still L2-hot — a separate CRC pass re-reads
// the whole file image from HBM (~0.5 ms / 215 MB, r01 profile).
__global__ void emit_records_kernel(const FieldColumn* __restrict__ cols,
                                    const u8* __restrict__ schema_blob, int32_t fmt,
                                    i64 r0, i64 R, const i64* __restrict__ frame_off,
                                    u8* __restrict__ file,
                                    int32_t* __restrict__ err) {
  __shared__ uint32_t tab[8][256];
  stage_crc_tables(tab);
  SchemaView schema = schema_view(schema_blob);
  for (i64 r = r0 + blockIdx.x * (i64)blockDim.x + threadIdx.x; r < R;
       r += (i64)gridDim.x * blockDim.x) {
    i64 payload = (frame_off[r + 1] - frame_off[r]) - kFrameOverhead;
    u8* o = file + frame_off[r] + 12;
    u32 crc = 0;
    i64 emitted = emit_record_payload_fused(o, cols, schema, fmt, r, &crc,
                                            tab);
    if (emitted != payload) set_err(err, ERR_OVERFLOW, r);
    write_frame_header_footer_crc(file, frame_off[r], payload, crc, tab);
  }
}

// ---------------------------------------------------------------------------
// Wavefront-cooperative large records. One-lane-per-record starves the chip
// when records are big and few (16k x 16 KB records measured 13 GB/s vs
// 21 GB/s for small records): a whole wave owns one record instead — copies
// are lane-strided u64s, and the CRC splits into 64 contiguous chunks whose
// finalized CRCs fold left-to-right with a precomputed GF(2) shift operator
// (crc32c_shift_op; one ~log2(chunk) matrix build per record on lane 0).
// ---------------------------------------------------------------------------

constexpr i64 kWaveRecordBytes = 8 << 10;  // avg record size to switch modes

__device__ inline u32 crc32c_wave(const u8* p, i64 n,
                                  const uint32_t (*tab)[256]) {
  int lane = threadIdx.x & 63;
  i64 chunk = (n + 63) / 64;
  if (chunk < 64) chunk = 64;  // tiny records: lane 0 does it all
  i64 s = (i64)lane * chunk;
  i64 e = s + chunk < n ? s + chunk : n;
  u32 my = (s < e) ? crc32c_sw(p + s, (size_t)(e - s), 0, tab) : 0;
  u32 acc = __shfl(my, 0);
  if (chunk < n) {
    // every lane folds redundantly in lockstep (uniform control flow, no
    // divergence around the cross-lane reads); acc ends identical
    // wave-wide. The fold is the register-only GF(2^32) field form —
    // the CrcMat version kept 128 B matrices in per-lane SCRATCH and made
    // this fold 30x more expensive than the chunk CRCs themselves.
    u32 op = crc32c_shift_elem((u64)chunk);
    for (int i = 1; i < 64; ++i) {
      i64 si = (i64)i * chunk;
      if (si >= n) break;
      u32 ci = __shfl(my, i);
      i64 li = (si + chunk < n ? chunk : n - si);
      acc = (li == chunk) ? (crc_gfmul(acc, op) ^ ci)
                          : crc32c_combine_fast(acc, ci, (u64)li);
    }
  }
  return acc;
}

__device__ inline void copy_wave(u8* d, const u8* s, i64 n) {
  int lane = threadIdx.x & 63;
  i64 i = (i64)lane * 8;
  for (; i + 8 <= n; i += 64 * 8) {
    u64 w;
    __builtin_memcpy(&w, s + i, 8);
    __builtin_memcpy(d + i, &w, 8);
  }
  i64 tail = n & ~((i64)7);
  for (i64 b = tail + lane; b < n; b += 64) d[b] = s[b];
}

// One record per WAVE: verify header + payload CRCs of large records.
__global__ void crc_verify_wave_kernel(const u8* __restrict__ data,
                                       const i64* __restrict__ off,
                                       const i64* __restrict__ len, i64 R,
                                       unsigned long long* err_out) {
  __shared__ uint32_t tab[8][256];
  stage_crc_tables(tab);
  i64 wave = (blockIdx.x * (i64)blockDim.x + threadIdx.x) / 64;
  i64 nwaves = ((i64)gridDim.x * blockDim.x) / 64;
  for (i64 r = wave; r < R; r += nwaves) {
    const u8* h = data + off[r] - 12;
    u32 payload = crc32c_wave(h + 12, len[r], tab);
    if ((threadIdx.x & 63) == 0) {
      u32 len_crc, data_crc;
      __builtin_memcpy(&len_crc, h + 8, 4);
      __builtin_memcpy(&data_crc, h + 12 + len[r], 4);
      bool ok = mask_crc(crc32c_sw(h, 8, 0, tab)) == len_crc &&
                mask_crc(payload) == data_crc;
      if (!ok) atomicMin(err_out, (unsigned long long)(r + 1));
    }
  }
}

// One record per WAVE: ByteArray framing of large payloads.
__global__ void frame_bytes_wave_kernel(const u8* __restrict__ src,
                                        const i64* __restrict__ elem_off,
                                        const i64* __restrict__ frame_off, i64 R,
                                        u8* __restrict__ file) {
  __shared__ uint32_t tab[8][256];
  stage_crc_tables(tab);
  i64 wave = (blockIdx.x * (i64)blockDim.x + threadIdx.x) / 64;
  i64 nwaves = ((i64)gridDim.x * blockDim.x) / 64;
  for (i64 r = wave; r < R; r += nwaves) {
    i64 n = elem_off[r + 1] - elem_off[r];
    const u8* s = src + elem_off[r];
    u8* d = file + frame_off[r] + 12;
    copy_wave(d, s, n);
    u32 crc = crc32c_wave(s, n, tab);
    if ((threadIdx.x & 63) == 0)
      write_frame_header_footer_crc(file, frame_off[r], n, crc, tab);
  }
}

// One record per WAVE: payload gather of large extents.
__global__ void gather_payloads_wave_kernel(const u8* __restrict__ data,
                                            const i64* __restrict__ off,
                                            const i64* __restrict__ len,
                                            const i64* __restrict__ dst_off,
                                            i64 R, u8* __restrict__ out) {
  i64 wave = (blockIdx.x * (i64)blockDim.x + threadIdx.x) / 64;
  i64 nwaves = ((i64)gridDim.x * blockDim.x) / 64;
  for (i64 r = wave; r < R; r += nwaves)
    copy_wave(out + dst_off[r], data + off[r], len[r]);
}

// ---------------------------------------------------------------------------
// Wave-cooperative PROTO decode/encode for large records (roadmap r1 #5).
// One-lane-per-record starves the chip on big records because every payload
// byte funnels through one lane's serial CRC/copy loops. For records above
// kWaveRecordBytes a whole wave owns one record:
//  - structure walk (tags, lens, keys): ALL lanes execute it in lockstep —
//    the loads are wave-uniform (same address), so they broadcast from one
//    cache-line fetch, and keeping control flow uniform lets the walk call
//    wave-cooperative helpers at the value runs;
//  - value runs, which carry ~all the bytes of a big record, are processed
//    by all 64 lanes: packed floats / string bytes via lane-strided u64
//    copies, packed int64 via a parallel varint decode (per-lane terminator
//    counts + a shfl exclusive scan give each lane its output slots);
//  - the frame CRC is crc32c_wave (64 chunk CRCs folded with the GF(2)
//    shift operator) instead of the one-lane fused cursor.
// ---------------------------------------------------------------------------

// Inclusive shfl scan; returns this lane's inclusive prefix of v.
__device__ inline i64 wave_incl_scan(i64 v) {
  int lane = threadIdx.x & 63;
  for (int d = 1; d < 64; d <<= 1) {
    i64 y = __shfl_up(v, d);
    if (lane >= d) v += y;
  }
  return v;
}

// Parallel decode of one packed-int64 run [p, p+n) into out[base...].
// Varint starts are positions q with q==0 or MSB-clear at q-1; each lane
// owns the values STARTING in its contiguous chunk (a varint may spill into
// the next lane's chunk — reads cross, ownership doesn't). Returns the
// total value count (uniform across the wave).
__device__ inline i64 wave_extract_i64_packed(const u8* __restrict__ p, i64 n,
                                              i64* __restrict__ out, i64 base) {
  int lane = threadIdx.x & 63;
  i64 chunk = (n + 63) / 64;
  i64 lo = (i64)lane * chunk;
  i64 hi = lo + chunk < n ? lo + chunk : n;
  i64 cnt = 0;
  for (i64 i = lo; i < hi; ++i) cnt += !(p[i] & 0x80);
  i64 incl = wave_incl_scan(cnt);
  i64 excl = incl - cnt;
  i64 total = __shfl(incl, 63);
  // first varint start in my chunk (skip a varint spanning in from the left)
  i64 q = lo;
  i64 pre = 0;
  if (lo > 0 && lo < hi) {
    if (p[lo - 1] & 0x80) {
      while (q < hi && (p[q] & 0x80)) ++q;
      if (q < hi) {
        ++q;  // consumed the spanning varint's terminator
        pre = 1;
      }
    }
  }
  i64 idx = base + excl + pre;
  while (q < hi) {
    u64 v = 0;
    int shift = 0;
    i64 j = q;
    while (j < n && shift < 70) {
      u8 b = p[j++];
      v |= (u64)(b & 0x7F) << shift;
      shift += 7;
      if (!(b & 0x80)) break;
    }
    out[idx++] = (i64)v;
    q = j;
  }
  return total;
}

// Wave variant of extract_list_body: all lanes walk in lockstep, value runs
// go cooperative. vi/bi stay register-uniform (every lane computes the same
// updates).
__device__ inline int32_t extract_list_body_wave(const u8* __restrict__ p,
                                                 const u8* end, int32_t kind,
                                                 const DecodeDst& dst, i64* vi,
                                                 i64* bi) {
  int lane = threadIdx.x & 63;
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) return ERR_BAD_VARINT;
    u32 fieldno = (u32)(tag >> 3), wt = (u32)(tag & 7);
    if (fieldno != 1) {
      p = skip_field(p, end, wt);
      if (!p) return ERR_TRUNCATED;
      continue;
    }
    if (kind == KIND_BYTES) {
      u64 len;
      p = read_varint(p, end, &len);
      if (!p || (u64)(end - p) < len) return ERR_TRUNCATED;
      copy_wave(dst.bytes_data + *bi, p, (i64)len);
      if (lane == 0) dst.elem_len[*vi] = (i64)len;
      *bi += (i64)len;
      *vi += 1;
      p += len;
    } else if (kind == KIND_FLOAT) {
      if (wt == 2) {
        u64 len;
        p = read_varint(p, end, &len);
        if (!p || (u64)(end - p) < len) return ERR_TRUNCATED;
        u64 nv = len / 4;
        copy_wave((u8*)(dst.f32_vals + *vi), p, (i64)(nv * 4));
        *vi += (i64)nv;
        p += len;
      } else {
        if (end - p < 4) return ERR_TRUNCATED;
        if (lane == 0) {
          float v;
          __builtin_memcpy(&v, p, 4);
          dst.f32_vals[*vi] = v;
        }
        (*vi)++;
        p += 4;
      }
    } else {  // INT64
      if (wt == 2) {
        u64 len;
        p = read_varint(p, end, &len);
        if (!p || (u64)(end - p) < len) return ERR_TRUNCATED;
        *vi += wave_extract_i64_packed(p, (i64)len, dst.i64_vals, *vi);
        p += len;
      } else {
        u64 v;
        p = read_varint(p, end, &v);
        if (!p) return ERR_BAD_VARINT;
        if (lane == 0) dst.i64_vals[*vi] = (i64)v;
        (*vi)++;
      }
    }
  }
  return ERR_OK;
}

__device__ inline int32_t extract_feature_body_wave(const u8* __restrict__ p,
                                                    const u8* end, int32_t kind,
                                                    const DecodeDst& dst,
                                                    i64* vi, i64* bi) {
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) return ERR_BAD_VARINT;
    u32 fieldno = (u32)(tag >> 3), wt = (u32)(tag & 7);
    if (fieldno >= 1 && fieldno <= 3 && wt == 2) {
      u64 len;
      p = read_varint(p, end, &len);
      if (!p || (u64)(end - p) < len) return ERR_TRUNCATED;
      int32_t rc = extract_list_body_wave(p, p + len, kind, dst, vi, bi);
      if (rc != ERR_OK) return rc;
      p += len;
    } else {
      p = skip_field(p, end, wt);
      if (!p) return ERR_TRUNCATED;
    }
  }
  return ERR_OK;
}

__device__ inline int32_t extract_field_wave(const u8* __restrict__ data,
                                             i64 pos, i64 len, int32_t kind,
                                             int32_t is_seq, const DecodeDst& dst,
                                             i64 val_base, i64 byte_base,
                                             i64 list_base) {
  if (pos < 0) return ERR_OK;
  int lane = threadIdx.x & 63;
  const u8* p = data + pos;
  const u8* end = p + len;
  i64 vi = val_base;
  i64 bi = byte_base;
  if (!is_seq) return extract_feature_body_wave(p, end, kind, dst, &vi, &bi);
  i64 li = list_base;
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) return ERR_BAD_VARINT;
    u32 fieldno = (u32)(tag >> 3), wt = (u32)(tag & 7);
    if (fieldno == 1 && wt == 2) {
      u64 flen;
      p = read_varint(p, end, &flen);
      if (!p || (u64)(end - p) < flen) return ERR_TRUNCATED;
      i64 v_before = vi;
      int32_t rc = extract_feature_body_wave(p, p + flen, kind, dst, &vi, &bi);
      if (rc != ERR_OK) return rc;
      if (lane == 0) dst.sub_count[li] = vi - v_before;
      ++li;
      p += flen;
    } else {
      p = skip_field(p, end, wt);
      if (!p) return ERR_TRUNCATED;
    }
  }
  return ERR_OK;
}

// One record per WAVE structure scan: lane 0 runs the (cheap, structure-only)
// two-pass scan; the payload CRC — the byte-proportional part — runs
// wave-cooperatively. Used when records are large and few.
__global__ void scan_records_wave_kernel(const u8* __restrict__ data,
                                         const i64* __restrict__ off,
                                         const i64* __restrict__ len, i64 R,
                                         int32_t fmt,
                                         const u8* __restrict__ schema_blob,
                                         int F, FieldStat* __restrict__ stats,
                                         int32_t* __restrict__ err,
                                         unsigned long long* crc_err) {
  __shared__ uint32_t tab[8][256];
  stage_crc_tables(tab);
  SchemaView schema = schema_view(schema_blob);
  i64 wave = (blockIdx.x * (i64)blockDim.x + threadIdx.x) / 64;
  i64 nwaves = ((i64)gridDim.x * blockDim.x) / 64;
  int lane = threadIdx.x & 63;
  for (i64 r = wave; r < R; r += nwaves) {
    if (lane == 0) {
      FieldStat* st = stats + r * F;
      for (int f = 0; f < F; ++f) field_stat_clear(&st[f]);
      int32_t rc = scan_record(data, off[r], len[r], fmt, schema, st);
      if (rc != ERR_OK) set_err(err, rc, r);
      for (int f = 0; f < F; ++f)
        if (st[f].err != ERR_OK) set_err(err, st[f].err, r);
    }
    if (crc_err) {
      const u8* h = data + off[r] - 12;
      u32 payload = crc32c_wave(h + 12, len[r], tab);
      if (lane == 0) {
        u32 len_crc, data_crc;
        __builtin_memcpy(&len_crc, h + 8, 4);
        __builtin_memcpy(&data_crc, h + 12 + len[r], 4);
        bool ok = mask_crc(crc32c_sw(h, 8, 0, tab)) == len_crc &&
                  mask_crc(payload) == data_crc;
        if (!ok) atomicMin(crc_err, (unsigned long long)(r + 1));
      }
    }
  }
}

// One (record, field) pair per WAVE value extraction.
__global__ void extract_fields_wave_kernel(const u8* __restrict__ data, i64 R,
                                           int F,
                                           const FieldStat* __restrict__ stats,
                                           const DevFieldDst* __restrict__ metas,
                                           int32_t* __restrict__ err) {
  i64 total = R * F;
  i64 wave = (blockIdx.x * (i64)blockDim.x + threadIdx.x) / 64;
  i64 nwaves = ((i64)gridDim.x * blockDim.x) / 64;
  for (i64 idx = wave; idx < total; idx += nwaves) {
    i64 r = idx / F;
    int f = (int)(idx - r * F);
    const DevFieldDst& m = metas[f];
    const FieldStat& st = stats[r * F + f];
    DecodeDst dst{m.i64_vals, m.f32_vals, m.bytes_data, m.elem_len, m.sub_count};
    int32_t rc = extract_field_wave(data, st.pos, st.len, m.kind, m.is_seq, dst,
                                    m.val_base ? m.val_base[r] : 0,
                                    m.byte_base ? m.byte_base[r] : 0,
                                    m.list_base ? m.list_base[r] : 0);
    if (rc != ERR_OK && (threadIdx.x & 63) == 0) set_err(err, rc, r);
  }
}

// ---------------------------------------------------------------------------
// Wave-cooperative EMIT for large records. Structure bytes are stored by
// lane 0 (all lanes track the cursor in registers); value runs are emitted
// by the whole wave; the frame CRC re-reads the emitted payload wave-wide
// (L2-hot) instead of the one-lane fused WriteCur.
// ---------------------------------------------------------------------------

// Emits a packed-int64 run's varints in parallel: lane l takes values
// [v0 + l*chunk ...), sums its varint sizes, a shfl scan places its output
// cursor, then it emits serially. Returns total packed bytes (uniform).
__device__ inline i64 wave_emit_i64_packed(u8* __restrict__ o,
                                           const i64* __restrict__ vals,
                                           i64 v0, i64 v1) {
  int lane = threadIdx.x & 63;
  i64 n = v1 - v0;
  i64 chunk = (n + 63) / 64;
  i64 lo = v0 + (i64)lane * chunk;
  i64 hi = lo + chunk < v1 ? lo + chunk : v1;
  i64 sz = 0;
  for (i64 v = lo; v < hi; ++v) sz += varint_size((u64)vals[v]);
  i64 incl = wave_incl_scan(sz);
  u8* q = o + (incl - sz);
  for (i64 v = lo; v < hi; ++v) q = write_varint(q, (u64)vals[v]);
  return __shfl(incl, 63);
}

struct WaveCur {
  u8* base;
  i64 pos;  // uniform across the wave
};

__device__ inline void wvc_put(WaveCur& w, u8 b) {
  if ((threadIdx.x & 63) == 0) w.base[w.pos] = b;
  w.pos += 1;
}

__device__ inline void wvc_varint(WaveCur& w, u64 v) {
  if ((threadIdx.x & 63) == 0) write_varint(w.base + w.pos, v);
  w.pos += varint_size(v);
}

__device__ inline void wvc_bytes(WaveCur& w, const u8* s, i64 n) {
  copy_wave(w.base + w.pos, s, n);
  w.pos += n;
}

__device__ inline void wvc_list_body(WaveCur& w, const FieldColumn& c, i64 v0,
                                     i64 v1) {
  if (c.kind == KIND_FLOAT) {
    i64 n = v1 - v0;
    if (n) {
      wvc_put(w, 0x0A);
      wvc_varint(w, (u64)(4 * n));
      wvc_bytes(w, (const u8*)(c.f32_vals + v0), 4 * n);
    }
  } else if (c.kind == KIND_INT64) {
    i64 packed = 0;
    for (i64 v = v0; v < v1; ++v) packed += varint_size((u64)c.i64_vals[v]);
    if (packed) {
      wvc_put(w, 0x0A);
      wvc_varint(w, (u64)packed);
      w.pos += wave_emit_i64_packed(w.base + w.pos, c.i64_vals, v0, v1);
    }
  } else {
    for (i64 v = v0; v < v1; ++v) {
      i64 b0 = c.elem_off[v];
      i64 blen = c.elem_off[v + 1] - b0;
      wvc_put(w, 0x0A);
      wvc_varint(w, (u64)blen);
      wvc_bytes(w, c.bytes_data + b0, blen);
    }
  }
}

__device__ inline void wvc_feature_body(WaveCur& w, const FieldColumn& c,
                                        i64 v0, i64 v1) {
  i64 body = list_body_size(c, v0, v1);
  wvc_put(w, (u8)((c.kind << 3) | 2));
  wvc_varint(w, (u64)body);
  wvc_list_body(w, c, v0, v1);
}

__device__ inline void wvc_features_entry(WaveCur& w, const FieldColumn& c,
                                          const SchemaView& s, int f, i64 v0,
                                          i64 v1) {
  i64 fb = feature_body_size(c, v0, v1);
  i64 klen = s.name_len(f);
  wvc_put(w, 0x0A);
  wvc_varint(w, (u64)klen);
  wvc_bytes(w, s.name(f), klen);
  wvc_put(w, 0x12);
  wvc_varint(w, (u64)fb);
  wvc_feature_body(w, c, v0, v1);
}

__device__ inline void wvc_feature_lists_entry(WaveCur& w, const FieldColumn& c,
                                               const SchemaView& s, int f,
                                               i64 r) {
  i64 flb = feature_list_body_size(c, r);
  i64 klen = s.name_len(f);
  wvc_put(w, 0x0A);
  wvc_varint(w, (u64)klen);
  wvc_bytes(w, s.name(f), klen);
  wvc_put(w, 0x12);
  wvc_varint(w, (u64)flb);
  for (i64 j = c.list_off[r]; j < c.list_off[r + 1]; ++j) {
    i64 fb = feature_body_size(c, c.sub_off[j], c.sub_off[j + 1]);
    wvc_put(w, 0x0A);
    wvc_varint(w, (u64)fb);
    wvc_feature_body(w, c, c.sub_off[j], c.sub_off[j + 1]);
  }
}

__device__ inline i64 emit_record_payload_wave(u8* o, const FieldColumn* cols,
                                               const SchemaView& s, int32_t fmt,
                                               i64 r) {
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
  WaveCur w{o, 0};
  auto emit_ctx = [&]() {
    for (int f = 0; f < s.nfields; ++f) {
      const FieldColumn& c = cols[f];
      if (!c.presence[r] || c.is_seq) continue;
      i64 e = features_entry_size(c, s, f, c.row_off[r], c.row_off[r + 1]);
      wvc_put(w, 0x0A);
      wvc_varint(w, (u64)e);
      wvc_features_entry(w, c, s, f, c.row_off[r], c.row_off[r + 1]);
    }
  };
  auto emit_fl = [&]() {
    for (int f = 0; f < s.nfields; ++f) {
      const FieldColumn& c = cols[f];
      if (!c.presence[r] || !c.is_seq) continue;
      i64 e = feature_lists_entry_size(c, s, f, r);
      wvc_put(w, 0x0A);
      wvc_varint(w, (u64)e);
      wvc_feature_lists_entry(w, c, s, f, r);
    }
  };
  if (fmt == FMT_EXAMPLE) {
    wvc_put(w, 0x0A);
    wvc_varint(w, (u64)ctx_body);
    emit_ctx();
  } else {
    if (ctx_body) {
      wvc_put(w, 0x0A);
      wvc_varint(w, (u64)ctx_body);
      emit_ctx();
    }
    wvc_put(w, 0x12);
    wvc_varint(w, (u64)fl_body);
    emit_fl();
  }
  return w.pos;
}

// One record per WAVE emit: payload cooperatively, then wave CRC of the
// freshly-stored (L2-hot) payload, then lane 0 writes the frame header.
__global__ void emit_records_wave_kernel(const FieldColumn* __restrict__ cols,
                                         const u8* __restrict__ schema_blob,
                                         int32_t fmt, i64 r0, i64 R,
                                         const i64* __restrict__ frame_off,
                                         u8* __restrict__ file,
                                         int32_t* __restrict__ err) {
  __shared__ uint32_t tab[8][256];
  stage_crc_tables(tab);
  SchemaView schema = schema_view(schema_blob);
  i64 wave = (blockIdx.x * (i64)blockDim.x + threadIdx.x) / 64;
  i64 nwaves = ((i64)gridDim.x * blockDim.x) / 64;
  int lane = threadIdx.x & 63;
  for (i64 r = r0 + wave; r < R; r += nwaves) {
    i64 payload = (frame_off[r + 1] - frame_off[r]) - kFrameOverhead;
    u8* o = file + frame_off[r] + 12;
    i64 emitted = emit_record_payload_wave(o, cols, schema, fmt, r);
    if (emitted != payload && lane == 0) set_err(err, ERR_OVERFLOW, r);
    // make every lane's payload stores visible before other lanes re-read
    // them for the CRC (lockstep wave => the fence alone is the barrier)
    __threadfence_block();
    u32 crc = crc32c_wave(o, payload, tab);
    if (lane == 0)
      write_frame_header_footer_crc(file, frame_off[r], payload, crc, tab);
  }
}

// ByteArray framing: payload r occupies [elem_off[r], elem_off[r+1]) of src.
// The copy streams through a WriteCur so the frame CRC comes from the same
// register windows the stores use (no payload re-read).
__global__ void frame_bytes_kernel(const u8* __restrict__ src,
                                   const i64* __restrict__ elem_off,
                                   const i64* __restrict__ frame_off, i64 R,
                                   u8* __restrict__ file) {
  __shared__ uint32_t tab[8][256];
  stage_crc_tables(tab);
  for (i64 r = blockIdx.x * (i64)blockDim.x + threadIdx.x; r < R;
       r += (i64)gridDim.x * blockDim.x) {
    i64 n = elem_off[r + 1] - elem_off[r];
    WriteCur w;
    wcur_init(w, file + frame_off[r] + 12, tab);
    wcur_bytes(w, src + elem_off[r], n);
    u32 crc = wcur_finish(w);
    write_frame_header_footer_crc(file, frame_off[r], n, crc, tab);
  }
}

// Parallel frame-boundary discovery (SURVEY.md §7 "hard parts"): TFRecord has
// no sync markers, so every byte position is tested as a candidate frame head
// (8-byte length whose masked CRC32C matches the next 4 bytes). A real head's
// 12-byte check passes always; a random position passes with p = 2^-32, so
// candidates ~= frames. Python sorts the candidates and validates the chain
// pos[k+1] == pos[k] + 16 + len[k] with two tensor ops — no host pass over
// the file bytes at all.
// CRC32C of 8 little-endian bytes held in a register (one slicing-by-8 step).
__device__ inline u32 crc8b_reg(u64 w, const uint32_t (*tab)[256]) {
  w ^= 0xFFFFFFFFull;  // crc init (~0) folded into low 32 bits
  u32 c = tab[7][w & 0xFF] ^ tab[6][(w >> 8) & 0xFF] ^ tab[5][(w >> 16) & 0xFF] ^
          tab[4][(w >> 24) & 0xFF] ^ tab[3][(w >> 32) & 0xFF] ^
          tab[2][(w >> 40) & 0xFF] ^ tab[1][(w >> 48) & 0xFF] ^
          tab[0][(w >> 56) & 0xFF];
  return ~c;
}

// Each lane owns kPosPerLane consecutive byte positions; the covering bytes
// are loaded once as aligned u64 words and every position's (length, crc)
// pair is reassembled with 128-bit funnel shifts — no per-byte loads. The
// length bound check (random u64 <= N with p ~ N/2^64) culls everything but
// real frame heads before any CRC work.
//
// Deterministic two-pass compaction: pass 1 (COUNT) writes per-block
// candidate counts, Python prefix-sums them, pass 2 (EMIT) re-evaluates and
// writes candidates at exact offsets — output is SORTED BY POSITION by
// construction (block b covers chunks [c_lo + b*blockDim, ...); thread t its
// t-th chunk; positions ascend within a chunk), so no device sort and no
// contended atomics at all. (The first design used per-candidate atomics on
// one global counter — ~11 ms of L2 serialization at 1M records, see
// profiles/r01_kernel_stats.txt — then an LDS-aggregated append that still
// needed a 0.9 ms device sort; this version needs neither.) The count pass
// runs per arrived slice under the H2D DMA of the pipelined reader.
constexpr int kPosPerLane = 16;

// Evaluates the (up to) 16 candidate positions of chunk c, calling
// f(pos, len) for each one that passes the length cull + header CRC.
template <typename F>
__device__ inline void eval_frame_chunk(const u8* __restrict__ data,
                                        const u64* __restrict__ wdata, i64 N,
                                        i64 c, i64 pos_start, i64 pos_end,
                                        const uint32_t (*tab)[256], F&& f) {
  // p0 = 16*c is 8-byte aligned, so every index below is compile-time
  // constant after unrolling (runtime-indexed w[] would spill to scratch).
  i64 p0 = c * kPosPerLane;
  i64 w0 = p0 >> 3;
  u64 w[4];
  if (p0 + 32 <= N) {  // interior chunk: full-word loads
#pragma unroll
    for (int j = 0; j < 4; ++j) w[j] = wdata[w0 + j];
  } else {  // file tail: assemble partial words byte-wise, zero-padded
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      u64 v = 0;
      i64 boff = (w0 + j) * 8;
      for (int b = 0; b < 8 && boff + b < N; ++b)
        v |= (u64)data[boff + b] << (8 * b);
      w[j] = v;
    }
  }
#pragma unroll
  for (int k = 0; k < kPosPerLane; ++k) {
    i64 i = p0 + k;
    if (i + 16 > N || i >= pos_end) break;
    if (i < pos_start) continue;
    constexpr int _ppl = kPosPerLane;
    static_assert(_ppl == 16, "index math below assumes 16 positions");
    const int wi = k >> 3;
    const int sh = (k & 7) * 8;
    u64 len = (sh == 0) ? w[wi] : (w[wi] >> sh) | (w[wi + 1] << (64 - sh));
    if (len > (u64)(N - i) - 16) continue;
    const int wi2 = (k + 8) >> 3;
    const int sh2 = sh;  // (k+8) & 7 == k & 7
    u64 hi = (sh2 == 0) ? w[wi2]
                        : (w[wi2] >> sh2) | (w[wi2 + 1] << (64 - sh2));
    u32 want = (u32)hi;
    if (mask_crc(crc8b_reg(len, tab)) != want) continue;
    f(i, (i64)len);
  }
}

// EMIT=false: block_data[block_base + b] = candidate count of block b.
// EMIT=true:  block_data holds the exclusive scan of those counts (read);
//             candidates land at cand_pos/cand_len[block offset + rank].
template <bool EMIT>
__global__ void frame_scan_pass_kernel(const u8* __restrict__ data, i64 N,
                                       i64 pos_start, i64 pos_end,
                                       i64 block_base,
                                       i64* __restrict__ block_data,
                                       i64* __restrict__ cand_pos,
                                       i64* __restrict__ cand_len) {
  __shared__ uint32_t tab[8][256];
  __shared__ unsigned int cnt[256];
  stage_crc_tables(tab);
  const u64* wdata = reinterpret_cast<const u64*>(data);  // data is 8B-aligned
  i64 c_lo = pos_start / kPosPerLane;
  i64 c_hi = (pos_end + kPosPerLane - 1) / kPosPerLane;
  i64 c = c_lo + (i64)blockIdx.x * blockDim.x + threadIdx.x;
  unsigned int my = 0;
  if (c < c_hi)
    eval_frame_chunk(data, wdata, N, c, pos_start, pos_end, tab,
                     [&](i64, i64) { ++my; });
  cnt[threadIdx.x] = my;
  __syncthreads();
  if (!EMIT) {
    for (int off = blockDim.x / 2; off; off >>= 1) {  // tree reduction
      if ((int)threadIdx.x < off) cnt[threadIdx.x] += cnt[threadIdx.x + off];
      __syncthreads();
    }
    if (threadIdx.x == 0) block_data[block_base + blockIdx.x] = (i64)cnt[0];
    return;
  }
  // block-local inclusive scan of per-thread counts (Hillis-Steele in LDS)
  for (int off = 1; off < (int)blockDim.x; off <<= 1) {
    unsigned int v = (int)threadIdx.x >= off ? cnt[threadIdx.x - off] : 0;
    __syncthreads();
    cnt[threadIdx.x] += v;
    __syncthreads();
  }
  i64 slot = block_data[block_base + blockIdx.x] + (i64)(cnt[threadIdx.x] - my);
  if (c < c_hi)
    eval_frame_chunk(data, wdata, N, c, pos_start, pos_end, tab,
                     [&](i64 i, i64 len) {
                       cand_pos[slot] = i;
                       cand_len[slot] = len;
                       ++slot;
                     });
}

// ---------------------------------------------------------------------------
// Schema inference on device ("per-GPU type-lattice histogram
// kernel + RCCL all-reduce"). One record per lane walks the Features /
// FeatureLists maps; feature names are interned into a global open-addressed
// hash table (FNV-1a 64, atomicCAS claim), lattice codes merge with
// atomicMax — the same commutative max the reference's RDD aggregate uses
// (TensorFlowInferSchema.scala:120-127). Slot layout is [hash, nameref,
// code] as 3 x i64; nameref packs (absolute name offset << 16 | name len)
// of the first occurrence so the host can recover the string. The table
// halves are context [0, S/2) and sequence [S/2, S) name spaces.
// ---------------------------------------------------------------------------

struct InferSlot {
  unsigned long long hash;  // 0 = empty
  long long nameref;
  long long code;
};
static_assert(sizeof(InferSlot) == 24, "layout shared with Python [S,3] i64");

__device__ inline u64 fnv1a64(const u8* s, u64 n) {
  u64 h = 1469598103934665603ull;
  for (u64 i = 0; i < n; ++i) h = (h ^ s[i]) * 1099511628211ull;
  return h ? h : 1ull;  // 0 is the empty-slot marker
}

// Per-block LDS staging table: nearly every record shares the same few
// feature names, so merging per record straight into the global table puts
// millions of contended atomics on a handful of L2 addresses (measured
// ~300k records/s). Each block merges into LDS first (fast, block-local)
// and flushes once at kernel end — global atomics drop to blocks x names.
constexpr int kLdsInfer = 64;

struct LdsInferSlot {
  unsigned long long hash;
  long long nameref;
  unsigned int code;
};

__device__ inline bool lds_infer_merge(LdsInferSlot* ltab, u64 hash,
                                       long long nameref, int code) {
  int idx = (int)(hash % (u64)kLdsInfer);
  for (int probe = 0; probe < kLdsInfer; ++probe) {
    LdsInferSlot& s = ltab[idx];
    unsigned long long seen =
        atomicCAS(&s.hash, 0ull, (unsigned long long)hash);
    if (seen == 0ull) s.nameref = nameref;
    if (seen == 0ull || seen == hash) {
      atomicMax(&s.code, (unsigned int)code);
      return true;
    }
    idx = (idx + 1) % kLdsInfer;
  }
  return false;  // >64 distinct names in this block: caller goes global
}

// Claims/finds the slot for (hash) in table half [lo, lo+half) and merges
// code with max. Returns false when the half is full (caller sets err).
__device__ inline bool infer_slot_merge(InferSlot* table, int lo, int half,
                                        u64 hash, long long nameref, int code) {
  int idx = lo + (int)(hash % (u64)half);
  for (int probe = 0; probe < half; ++probe) {
    InferSlot& s = table[idx];
    unsigned long long seen =
        atomicCAS(&s.hash, 0ull, (unsigned long long)hash);
    if (seen == 0ull) s.nameref = nameref;  // claimer records first occurrence
    if (seen == 0ull || seen == hash) {
      atomicMax((unsigned long long*)&s.code, (unsigned long long)code);
      return true;
    }
    idx = (idx + 1 - lo) % half + lo;
  }
  return false;
}

// Walks one Features / FeatureLists body, inferring per-entry lattice codes.
// Mirrors csrc/ext.cpp infer_features_body (host reference implementation).
__device__ inline int32_t infer_features_body_dev(const u8* data, const u8* p,
                                                  const u8* end, bool seq,
                                                  LdsInferSlot* ltab,
                                                  InferSlot* table, int nslots) {
  int half = nslots / 2;
  int lo = seq ? half : 0;
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) return ERR_BAD_VARINT;
    u32 fieldno = (u32)(tag >> 3), wt = (u32)(tag & 7);
    if (fieldno != 1 || wt != 2) {
      p = skip_field(p, end, wt);
      if (!p) return ERR_TRUNCATED;
      continue;
    }
    u64 entry_len;
    p = read_varint(p, end, &entry_len);
    if (!p || (u64)(end - p) < entry_len) return ERR_TRUNCATED;
    const u8* ep = p;
    const u8* ee = p + entry_len;
    p = ee;
    const u8* key = nullptr;
    u64 key_len = 0;
    const u8* val = nullptr;
    u64 val_len = 0;
    while (ep < ee) {
      u64 etag;
      ep = read_varint(ep, ee, &etag);
      if (!ep) return ERR_BAD_VARINT;
      u32 efn = (u32)(etag >> 3), ewt = (u32)(etag & 7);
      if (efn == 1 && ewt == 2) {
        ep = read_varint(ep, ee, &key_len);
        if (!ep || (u64)(ee - ep) < key_len) return ERR_TRUNCATED;
        key = ep;
        ep += key_len;
      } else if (efn == 2 && ewt == 2) {
        ep = read_varint(ep, ee, &val_len);
        if (!ep || (u64)(ee - ep) < val_len) return ERR_TRUNCATED;
        val = ep;
        ep += val_len;
      } else {
        ep = skip_field(ep, ee, ewt);
        if (!ep) return ERR_TRUNCATED;
      }
    }
    if (!key) continue;
    int code = 0;
    if (val) {
      if (!seq) {
        int32_t kf = 0;
        i64 nvals = 0, nbytes = 0;
        int32_t rc = scan_feature_body(val, val + val_len, -1, &kf, &nvals, &nbytes);
        if (rc != ERR_OK) return rc;
        // rank: int64->1 float->2 bytes->3; +3 when >1 element (array)
        if (kf != 0 && nvals > 0) code = (4 - kf) + (nvals > 1 ? 3 : 0);
      } else {
        const u8* lp = val;
        const u8* le = val + val_len;
        int kind_seen = 0;
        while (lp < le) {
          u64 ltag;
          lp = read_varint(lp, le, &ltag);
          if (!lp) return ERR_BAD_VARINT;
          u32 lfn = (u32)(ltag >> 3), lwt = (u32)(ltag & 7);
          if (lfn == 1 && lwt == 2) {
            u64 flen;
            lp = read_varint(lp, le, &flen);
            if (!lp || (u64)(le - lp) < flen) return ERR_TRUNCATED;
            int32_t kf = 0;
            i64 nvals = 0, nbytes = 0;
            int32_t rc = scan_feature_body(lp, lp + flen, -1, &kf, &nvals, &nbytes);
            if (rc != ERR_OK) return rc;
            if (kf) {
              int r = 4 - kf;
              kind_seen = kind_seen > r ? kind_seen : r;
            }
            lp += flen;
          } else {
            lp = skip_field(lp, le, lwt);
            if (!lp) return ERR_TRUNCATED;
          }
        }
        if (kind_seen) code = 6 + kind_seen;  // FeatureList infers 2-D
      }
    }
    if (key_len > 0xFFFF) return ERR_NAME_TOO_LONG;  // nameref packs len in 16 bits
    long long nameref = ((long long)(key - data) << 16) | (long long)key_len;
    u64 hash = fnv1a64(key, key_len);
    if (!lds_infer_merge(ltab, hash, nameref, code) &&
        !infer_slot_merge(table, lo, half, hash, nameref, code))
      return ERR_OVERFLOW;  // feature-name table full
  }
  return ERR_OK;
}

__global__ void infer_codes_kernel(const u8* __restrict__ data,
                                   const i64* __restrict__ off,
                                   const i64* __restrict__ len, i64 R, int32_t fmt,
                                   InferSlot* __restrict__ table, int nslots,
                                   int32_t* __restrict__ err) {
  __shared__ LdsInferSlot lctx[kLdsInfer];
  __shared__ LdsInferSlot lseq[kLdsInfer];
  for (int i = threadIdx.x; i < kLdsInfer; i += blockDim.x) {
    lctx[i].hash = 0;
    lctx[i].code = 0;
    lseq[i].hash = 0;
    lseq[i].code = 0;
  }
  __syncthreads();
  for (i64 r = blockIdx.x * (i64)blockDim.x + threadIdx.x; r < R;
       r += (i64)gridDim.x * blockDim.x) {
    const u8* p = data + off[r];
    const u8* end = p + len[r];
    while (p < end) {
      u64 tag;
      const u8* np = read_varint(p, end, &tag);
      if (!np) {
        set_err(err, ERR_BAD_VARINT, r);
        break;
      }
      p = np;
      u32 fieldno = (u32)(tag >> 3), wt = (u32)(tag & 7);
      bool is_features = fieldno == 1;
      bool is_fl = (fmt == FMT_SEQUENCE && fieldno == 2);
      if ((is_features || is_fl) && wt == 2) {
        u64 blen;
        p = read_varint(p, end, &blen);
        if (!p || (u64)(end - p) < blen) {
          set_err(err, ERR_TRUNCATED, r);
          break;
        }
        int32_t rc = infer_features_body_dev(data, p, p + blen, is_fl,
                                             is_fl ? lseq : lctx, table, nslots);
        if (rc != ERR_OK) {
          set_err(err, rc, r);
          break;
        }
        p += blen;
      } else {
        p = skip_field(p, end, wt);
        if (!p) {
          set_err(err, ERR_TRUNCATED, r);
          break;
        }
      }
    }
  }
  // flush the block-local tables into the global one (blocks x names atomics)
  __syncthreads();
  int half = nslots / 2;
  for (int i = threadIdx.x; i < kLdsInfer; i += blockDim.x) {
    if (lctx[i].hash &&
        !infer_slot_merge(table, 0, half, lctx[i].hash, lctx[i].nameref,
                          (int)lctx[i].code))
      err[0] = ERR_OVERFLOW;
    if (lseq[i].hash &&
        !infer_slot_merge(table, half, half, lseq[i].hash, lseq[i].nameref,
                          (int)lseq[i].code))
      err[0] = ERR_OVERFLOW;
  }
}

// Raw-payload gather (ByteArray reads, partitionBy framed-record gather):
// word-wise copies — gfx950 handles unaligned u64 loads/stores, so a byte
// loop would cost 8x the memory instructions.
__global__ void gather_payloads_kernel(const u8* __restrict__ data,
                                       const i64* __restrict__ off,
                                       const i64* __restrict__ len,
                                       const i64* __restrict__ dst_off, i64 R,
                                       u8* __restrict__ out) {
  for (i64 r = blockIdx.x * (i64)blockDim.x + threadIdx.x; r < R;
       r += (i64)gridDim.x * blockDim.x) {
    const u8* s = data + off[r];
    u8* d = out + dst_off[r];
    i64 n = len[r];
    i64 i = 0;
    for (; i + 8 <= n; i += 8) {
      u64 w;
      __builtin_memcpy(&w, s + i, 8);
      __builtin_memcpy(d + i, &w, 8);
    }
    for (; i < n; ++i) d[i] = s[i];
  }
}

// ---------------------------------------------------------------------------
// Launch wrappers (pointers are caller-owned torch tensors; stream is the
// torch current stream so the kernels interleave with cumsum/casts).
// ---------------------------------------------------------------------------

inline int grid_for(i64 n) {
  // ≫256 workgroups to fill 8 XCDs × 32 CUs; cap so grid-stride amortizes.
  i64 blocks = (n + kBlock - 1) / kBlock;
  if (blocks > 8192) blocks = 8192;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

void gpu_crc_verify(uintptr_t data, uintptr_t off, uintptr_t len, i64 R,
                    uintptr_t err_out, uintptr_t stream, i64 avg_bytes) {
  if (avg_bytes > kWaveRecordBytes) {  // few big records: one per wave
    hipLaunchKernelGGL(crc_verify_wave_kernel, dim3(grid_for(R * 64)),
                       dim3(kBlock), 0, (hipStream_t)stream, (const u8*)data,
                       (const i64*)off, (const i64*)len, R,
                       (unsigned long long*)err_out);
  } else {
    hipLaunchKernelGGL(crc_verify_kernel, dim3(grid_for(R)), dim3(kBlock), 0,
                       (hipStream_t)stream, (const u8*)data, (const i64*)off,
                       (const i64*)len, R, (unsigned long long*)err_out);
  }
  HIP_CHECK(hipGetLastError());
}

void gpu_scan_records(uintptr_t data, uintptr_t off, uintptr_t len, i64 R,
                      int32_t fmt, uintptr_t schema_blob, int F, uintptr_t stats,
                      uintptr_t err, uintptr_t crc_err, uintptr_t stream,
                      i64 avg_bytes) {
  if (avg_bytes > kWaveRecordBytes) {  // few big records: one per wave
    hipLaunchKernelGGL(scan_records_wave_kernel, dim3(grid_for(R * 64)),
                       dim3(kBlock), 0, (hipStream_t)stream, (const u8*)data,
                       (const i64*)off, (const i64*)len, R, fmt,
                       (const u8*)schema_blob, F, (FieldStat*)stats,
                       (int32_t*)err, (unsigned long long*)crc_err);
  } else {
    hipLaunchKernelGGL(scan_records_kernel, dim3(grid_for(R)), dim3(kBlock), 0,
                       (hipStream_t)stream, (const u8*)data, (const i64*)off,
                       (const i64*)len, R, fmt, (const u8*)schema_blob, F,
                       (FieldStat*)stats, (int32_t*)err,
                       (unsigned long long*)crc_err);
  }
  HIP_CHECK(hipGetLastError());
}

void gpu_extract_fields(uintptr_t data, i64 R, int F, uintptr_t stats,
                        py::list metas, uintptr_t meta_dev, uintptr_t err,
                        uintptr_t stream, i64 avg_bytes) {
  std::vector<DevFieldDst> host_metas(F);
  for (int f = 0; f < F; ++f) {
    py::dict d = metas[f].cast<py::dict>();
    DevFieldDst& m = host_metas[f];
    m.kind = d["kind"].cast<int32_t>();
    m.is_seq = d["is_seq"].cast<int32_t>();
    m.i64_vals = (i64*)d["i64_vals"].cast<uintptr_t>();
    m.f32_vals = (float*)d["f32_vals"].cast<uintptr_t>();
    m.bytes_data = (u8*)d["bytes_data"].cast<uintptr_t>();
    m.elem_len = (i64*)d["elem_len"].cast<uintptr_t>();
    m.sub_count = (i64*)d["sub_count"].cast<uintptr_t>();
    m.val_base = (const i64*)d["val_base"].cast<uintptr_t>();
    m.byte_base = (const i64*)d["byte_base"].cast<uintptr_t>();
    m.list_base = (const i64*)d["list_base"].cast<uintptr_t>();
  }
  HIP_CHECK(hipMemcpyAsync((void*)meta_dev, host_metas.data(),
                           sizeof(DevFieldDst) * F, hipMemcpyHostToDevice,
                           (hipStream_t)stream));
  if (avg_bytes > kWaveRecordBytes) {
    hipLaunchKernelGGL(extract_fields_wave_kernel, dim3(grid_for(R * F * 64)),
                       dim3(kBlock), 0, (hipStream_t)stream, (const u8*)data,
                       R, F, (const FieldStat*)stats,
                       (const DevFieldDst*)meta_dev, (int32_t*)err);
  } else {
    hipLaunchKernelGGL(extract_fields_kernel, dim3(grid_for(R * F)),
                       dim3(kBlock), 0, (hipStream_t)stream, (const u8*)data,
                       R, F, (const FieldStat*)stats,
                       (const DevFieldDst*)meta_dev, (int32_t*)err);
  }
  HIP_CHECK(hipGetLastError());
}

size_t gpu_devcols_bytes(int F) { return sizeof(FieldColumn) * (size_t)F; }

void fill_devcols(py::list cols, std::vector<FieldColumn>& out) {
  int F = (int)cols.size();
  out.resize(F);
  for (int f = 0; f < F; ++f) {
    py::dict d = cols[f].cast<py::dict>();
    FieldColumn& c = out[f];
    c.kind = d["kind"].cast<int32_t>();
    c.is_seq = d["is_seq"].cast<int32_t>();
    c.presence = (const u8*)d["presence"].cast<uintptr_t>();
    c.row_off = (const i64*)d["row_off"].cast<uintptr_t>();
    c.list_off = (const i64*)d["list_off"].cast<uintptr_t>();
    c.sub_off = (const i64*)d["sub_off"].cast<uintptr_t>();
    c.elem_off = (const i64*)d["elem_off"].cast<uintptr_t>();
    c.bytes_data = (const u8*)d["values_bytes"].cast<uintptr_t>();
    c.i64_vals = (const i64*)d["values_i64"].cast<uintptr_t>();
    c.f32_vals = (const float*)d["values_f32"].cast<uintptr_t>();
  }
}

void gpu_size_records(py::list cols, uintptr_t cols_dev, uintptr_t schema_blob,
                      int32_t fmt, i64 R, uintptr_t psize, uintptr_t stream) {
  // thread_local so the buffer outlives the async H2D enqueue (ROCm stages
  // pageable-source copies synchronously, but don't dangle a stack buffer)
  static thread_local std::vector<FieldColumn> host_cols;
  fill_devcols(cols, host_cols);
  HIP_CHECK(hipMemcpyAsync((void*)cols_dev, host_cols.data(),
                           sizeof(FieldColumn) * host_cols.size(),
                           hipMemcpyHostToDevice, (hipStream_t)stream));
  hipLaunchKernelGGL(size_records_kernel, dim3(grid_for(R)), dim3(kBlock), 0,
                     (hipStream_t)stream, (const FieldColumn*)cols_dev,
                     (const u8*)schema_blob, fmt, R, (i64*)psize);
  HIP_CHECK(hipGetLastError());
}

void gpu_emit_records(uintptr_t cols_dev, uintptr_t schema_blob, int32_t fmt,
                      i64 r0, i64 R, uintptr_t frame_off, uintptr_t file,
                      uintptr_t err, uintptr_t stream, i64 avg_bytes) {
  if (avg_bytes > kWaveRecordBytes) {
    hipLaunchKernelGGL(emit_records_wave_kernel, dim3(grid_for((R - r0) * 64)),
                       dim3(kBlock), 0, (hipStream_t)stream,
                       (const FieldColumn*)cols_dev, (const u8*)schema_blob,
                       fmt, r0, R, (const i64*)frame_off, (u8*)file,
                       (int32_t*)err);
  } else {
    hipLaunchKernelGGL(emit_records_kernel, dim3(grid_for(R - r0)),
                       dim3(kBlock), 0, (hipStream_t)stream,
                       (const FieldColumn*)cols_dev, (const u8*)schema_blob,
                       fmt, r0, R, (const i64*)frame_off, (u8*)file,
                       (int32_t*)err);
  }
  HIP_CHECK(hipGetLastError());
}

void gpu_frame_bytes(uintptr_t src, uintptr_t elem_off, uintptr_t frame_off, i64 R,
                     uintptr_t file, uintptr_t stream, i64 avg_bytes) {
  if (avg_bytes > kWaveRecordBytes) {
    hipLaunchKernelGGL(frame_bytes_wave_kernel, dim3(grid_for(R * 64)),
                       dim3(kBlock), 0, (hipStream_t)stream, (const u8*)src,
                       (const i64*)elem_off, (const i64*)frame_off, R,
                       (u8*)file);
  } else {
    hipLaunchKernelGGL(frame_bytes_kernel, dim3(grid_for(R)), dim3(kBlock), 0,
                       (hipStream_t)stream, (const u8*)src, (const i64*)elem_off,
                       (const i64*)frame_off, R, (u8*)file);
  }
  HIP_CHECK(hipGetLastError());
}

// Number of blocks a [pos_start, pos_end) range occupies in the frame-scan
// pass geometry (Python sizes the per-block count buffer with this).
i64 gpu_frame_scan_blocks(i64 pos_start, i64 pos_end) {
  i64 c_lo = pos_start / kPosPerLane;
  i64 c_hi = (pos_end + kPosPerLane - 1) / kPosPerLane;
  i64 n = c_hi - c_lo;
  return (n + kBlock - 1) / kBlock;
}

void gpu_frame_scan_count(uintptr_t data, i64 N, i64 pos_start, i64 pos_end,
                          i64 block_base, uintptr_t block_counts,
                          uintptr_t stream) {
  i64 B = gpu_frame_scan_blocks(pos_start, pos_end);
  if (B <= 0) return;
  hipLaunchKernelGGL(frame_scan_pass_kernel<false>, dim3((uint32_t)B),
                     dim3(kBlock), 0, (hipStream_t)stream, (const u8*)data, N,
                     pos_start, pos_end, block_base, (i64*)block_counts,
                     nullptr, nullptr);
  HIP_CHECK(hipGetLastError());
}

void gpu_frame_scan_emit(uintptr_t data, i64 N, i64 pos_start, i64 pos_end,
                         i64 block_base, uintptr_t block_off,
                         uintptr_t cand_pos, uintptr_t cand_len,
                         uintptr_t stream) {
  i64 B = gpu_frame_scan_blocks(pos_start, pos_end);
  if (B <= 0) return;
  hipLaunchKernelGGL(frame_scan_pass_kernel<true>, dim3((uint32_t)B),
                     dim3(kBlock), 0, (hipStream_t)stream, (const u8*)data, N,
                     pos_start, pos_end, block_base, (i64*)block_off,
                     (i64*)cand_pos, (i64*)cand_len);
  HIP_CHECK(hipGetLastError());
}

void gpu_infer_codes(uintptr_t data, uintptr_t off, uintptr_t len, i64 R,
                     int32_t fmt, uintptr_t table, int nslots, uintptr_t err,
                     uintptr_t stream) {
  hipLaunchKernelGGL(infer_codes_kernel, dim3(grid_for(R)), dim3(kBlock), 0,
                     (hipStream_t)stream, (const u8*)data, (const i64*)off,
                     (const i64*)len, R, fmt, (InferSlot*)table, nslots,
                     (int32_t*)err);
  HIP_CHECK(hipGetLastError());
}

void gpu_gather_payloads(uintptr_t data, uintptr_t off, uintptr_t len,
                         uintptr_t dst_off, i64 R, uintptr_t out,
                         uintptr_t stream, i64 avg_bytes) {
  if (avg_bytes > kWaveRecordBytes) {
    hipLaunchKernelGGL(gather_payloads_wave_kernel, dim3(grid_for(R * 64)),
                       dim3(kBlock), 0, (hipStream_t)stream, (const u8*)data,
                       (const i64*)off, (const i64*)len, (const i64*)dst_off,
                       R, (u8*)out);
  } else {
    hipLaunchKernelGGL(gather_payloads_kernel, dim3(grid_for(R)), dim3(kBlock),
                       0, (hipStream_t)stream, (const u8*)data,
                       (const i64*)off, (const i64*)len, (const i64*)dst_off,
                       R, (u8*)out);
  }
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// Prefix sums: rocprim decoupled-lookback scans. torch.cumsum's innermost-dim
// scan kernel launches one block per row — 2.5 ms for a [4, 1M] int64 scan
// (profiles/r01_kernel_stats.txt); these run at memory speed. The strided
// variant scans a column of the [R, F*6] FieldStat buffer in place, removing
// the transpose+contiguous materialization entirely.
// ---------------------------------------------------------------------------

struct StridedLoadI64 {
  const i64* p;
  i64 stride;
  __host__ __device__ i64 operator()(i64 i) const { return p[i * stride]; }
};

size_t gpu_scan_temp_bytes(i64 n) {
  size_t bytes = 0;
  auto it = rocprim::make_transform_iterator(
      rocprim::make_counting_iterator<i64>(0), StridedLoadI64{nullptr, 1});
  (void)rocprim::inclusive_scan(nullptr, bytes, it, (i64*)nullptr, (size_t)n,
                                rocprim::plus<i64>());
  return bytes;
}

// ---------------------------------------------------------------------------
// All per-field stat prefix sums in ONE launch: the decode needs exclusive
// scans of nvals/nbytes/nlists for EVERY field ([F][R+1] each). Issuing 3F
// rocprim scans costs 6F kernel launches; instead a single head-flag
// SEGMENTED scan runs over the virtual [F*R] sequence of (nvals, nbytes,
// nlists) triples (segments reset at field boundaries), scattering results
// into the three [F][R+1] output planes through a custom output iterator.
// ---------------------------------------------------------------------------

struct Seg3 {
  i64 v0, v1, v2;
  u32 flag;
};

struct Seg3Combine {
  __host__ __device__ Seg3 operator()(const Seg3& a, const Seg3& b) const {
    if (b.flag) return b;
    return Seg3{a.v0 + b.v0, a.v1 + b.v1, a.v2 + b.v2, a.flag};
  }
};

struct Seg3Load {
  const i64* stats;  // [R][F][6]
  i64 R;
  int F;
  __host__ __device__ Seg3 operator()(i64 idx) const {
    i64 f = idx / R;
    i64 r = idx - f * R;
    const i64* st = stats + (r * F + f) * 6;
    return Seg3{st[2], st[3], st[4], r == 0 ? 1u : 0u};
  }
};

// Writable proxy iterator: element idx scatters (v0,v1,v2) into the three
// output planes at [f][r+1] (out[f][0] is pre-zeroed by the caller).
struct Seg3OutRef {
  i64* o0;
  i64* o1;
  i64* o2;
  __host__ __device__ Seg3OutRef& operator=(const Seg3& v) {
    *o0 = v.v0;
    *o1 = v.v1;
    *o2 = v.v2;
    return *this;
  }
};

struct Seg3OutIt {
  using value_type = Seg3;
  using reference = Seg3OutRef;
  using pointer = Seg3OutRef*;
  using difference_type = i64;
  using iterator_category = std::random_access_iterator_tag;
  i64* b0;
  i64* b1;
  i64* b2;  // [F][R+1] planes
  i64 R;
  i64 pos;
  __host__ __device__ Seg3OutRef operator*() const { return (*this)[0]; }
  __host__ __device__ Seg3OutRef operator[](i64 k) const {
    i64 idx = pos + k;
    i64 f = idx / R;
    i64 r = idx - f * R;
    i64 at = f * (R + 1) + r + 1;
    return Seg3OutRef{b0 + at, b1 + at, b2 + at};
  }
  __host__ __device__ Seg3OutIt operator+(i64 k) const {
    return Seg3OutIt{b0, b1, b2, R, pos + k};
  }
  __host__ __device__ Seg3OutIt& operator+=(i64 k) {
    pos += k;
    return *this;
  }
  __host__ __device__ Seg3OutIt& operator++() {
    ++pos;
    return *this;
  }
};

size_t gpu_stat_scan_temp_bytes(i64 R, int F) {
  size_t bytes = 0;
  auto in = rocprim::make_transform_iterator(
      rocprim::make_counting_iterator<i64>(0), Seg3Load{nullptr, R, F});
  Seg3OutIt out{nullptr, nullptr, nullptr, R, 0};
  (void)rocprim::inclusive_scan(nullptr, bytes, in, out, (size_t)(R * F),
                                Seg3Combine());
  return bytes;
}

void gpu_stat_scans(uintptr_t temp, size_t temp_bytes, uintptr_t stats, i64 R,
                    int F, uintptr_t val_base, uintptr_t byte_base,
                    uintptr_t list_base, uintptr_t stream) {
  auto s = (hipStream_t)stream;
  if (R <= 0 || F <= 0) return;
  auto in = rocprim::make_transform_iterator(
      rocprim::make_counting_iterator<i64>(0),
      Seg3Load{(const i64*)stats, R, F});
  Seg3OutIt out{(i64*)val_base, (i64*)byte_base, (i64*)list_base, R, 0};
  size_t tb = temp_bytes;
  HIP_CHECK(rocprim::inclusive_scan((void*)temp, tb, in, out,
                                    (size_t)(R * F), Seg3Combine(), s));
}

// Writes out[0] = 0 and out[1..n] = inclusive scan of in[0], in[stride], ...
// i.e. out is the (n+1)-long exclusive scan with the total at out[n].
void gpu_excl_sum_strided(uintptr_t temp, size_t temp_bytes, uintptr_t in,
                          i64 stride, uintptr_t out, i64 n, uintptr_t stream) {
  auto s = (hipStream_t)stream;
  HIP_CHECK(hipMemsetAsync((void*)out, 0, sizeof(i64), s));
  if (n <= 0) return;
  auto it = rocprim::make_transform_iterator(
      rocprim::make_counting_iterator<i64>(0), StridedLoadI64{(const i64*)in, stride});
  size_t tb = temp_bytes;
  hipError_t e = rocprim::inclusive_scan((void*)temp, tb, it, (i64*)out + 1,
                                         (size_t)n, rocprim::plus<i64>(), s);
  HIP_CHECK(e);
}

// ---------------------------------------------------------------------------
// Pinned file mappings: mmap a tmpfs/page-cache file and hipHostRegister the
// mapping so hipMemcpyAsync can DMA directly between HBM and the file's own
// pages — no pinned bounce buffer, no host memcpy, no inode-locked write()
// (parallel pwrite to one tmpfs file serializes on the inode mutex; mapped
// DMA measured this at 27.7 ms -> ~4 ms for a 215 MB file image).
// Mappings are cached per path keyed by (dev, ino, size) so the registration
// cost amortizes across steps; an unlink+recreate changes the inode and
// evicts the stale entry.
// ---------------------------------------------------------------------------

struct MappedFile {
  void* ptr = nullptr;
  size_t n = 0;
  dev_t dev = 0;
  ino_t ino = 0;
  bool pinned = false;
  bool writable = false;
};

std::map<std::string, MappedFile> g_mmap_cache;
std::mutex g_mmap_mu;
size_t g_mmap_bytes = 0;
constexpr size_t kMmapCacheCap = 4ull << 30;  // 4 GiB of cached mappings (8 ranks share a host)

void drop_mapping_locked(const std::string& key) {
  auto it = g_mmap_cache.find(key);
  if (it == g_mmap_cache.end()) return;
  if (it->second.pinned) {
    (void)hipDeviceSynchronize();  // a DMA into this mapping may be in flight
    (void)hipHostUnregister(it->second.ptr);
  }
  munmap(it->second.ptr, it->second.n);
  g_mmap_bytes -= it->second.n;
  g_mmap_cache.erase(it);
}

// Drop cache entries whose path no longer resolves to the mapped inode
// (overwritten/renamed/deleted files): their pinned pages keep the DEAD
// tmpfs file alive until eviction otherwise. One stat per entry, entries
// are few.
void sweep_stale_locked() {
  for (auto it = g_mmap_cache.begin(); it != g_mmap_cache.end();) {
    struct stat st {};
    bool live = stat(it->first.c_str(), &st) == 0 &&
                st.st_dev == it->second.dev && st.st_ino == it->second.ino &&
                (size_t)st.st_size >= it->second.n;
    if (live) {
      ++it;
    } else {
      auto key = it->first;
      ++it;
      drop_mapping_locked(key);
    }
  }
}

// Returns (ptr, pinned). With writable=true the file is created/resized to n
// first. pinned=false means hipHostRegister failed (fall back to the staged
// pread/pwrite path); ptr is 0 iff n == 0.
py::tuple file_mmap_pinned(const std::string& path, i64 n, bool writable) {
  if (n <= 0) return py::make_tuple((uintptr_t)0, true);
  std::lock_guard<std::mutex> lk(g_mmap_mu);
  sweep_stale_locked();
  int flags = writable ? (O_RDWR | O_CREAT) : O_RDWR;
  int fd = ::open(path.c_str(), flags, 0644);
  bool rdonly = false;
  if (fd < 0 && !writable) {  // read path on a read-only file
    fd = ::open(path.c_str(), O_RDONLY);
    rdonly = true;
  }
  if (fd < 0)
    throw std::runtime_error("open failed: " + path + ": " + strerror(errno));
  struct stat st {};
  if (fstat(fd, &st) != 0) {
    ::close(fd);
    throw std::runtime_error("stat failed: " + path);
  }
  auto it = g_mmap_cache.find(path);
  if (it != g_mmap_cache.end()) {
    MappedFile& m = it->second;
    if (m.dev == st.st_dev && m.ino == st.st_ino && m.n == (size_t)n &&
        st.st_size >= (off_t)n && (m.writable || !writable)) {
      ::close(fd);
      return py::make_tuple((uintptr_t)m.ptr, m.pinned);
    }
    // drop BEFORE any resize: ftruncate below a pinned mapping blocks
    // forever on the DMA page pins
    drop_mapping_locked(path);
  }
  if (writable && st.st_size != (off_t)n && ftruncate(fd, (off_t)n) != 0) {
    int e = errno;
    ::close(fd);
    throw std::runtime_error("ftruncate failed: " + path + ": " + strerror(e));
  }
  if (!writable && st.st_size < (off_t)n) {
    ::close(fd);
    throw std::runtime_error("short file: " + path);
  }
  while (g_mmap_bytes + (size_t)n > kMmapCacheCap && !g_mmap_cache.empty())
    drop_mapping_locked(g_mmap_cache.begin()->first);
  int prot = PROT_READ | ((writable || !rdonly) ? PROT_WRITE : 0);
  void* p = mmap(nullptr, (size_t)n, prot, MAP_SHARED, fd, 0);
  ::close(fd);
  if (p == MAP_FAILED)
    throw std::runtime_error("mmap failed: " + path + ": " + strerror(errno));
  bool pinned = hipHostRegister(p, (size_t)n, hipHostRegisterDefault) == hipSuccess;
  if (!pinned) (void)hipGetLastError();  // clear sticky error
  MappedFile m{p, (size_t)n, st.st_dev, st.st_ino, pinned,
               writable || !rdonly};
  g_mmap_cache[path] = m;
  g_mmap_bytes += (size_t)n;
  return py::make_tuple((uintptr_t)p, pinned);
}

void file_mmap_drop(const std::string& path) {
  std::lock_guard<std::mutex> lk(g_mmap_mu);
  drop_mapping_locked(path);
}

// Flush a mapped file's pages to backing store before the caller publishes
// it by rename: DMA writes through a hipHostRegister'd mapping don't set
// CPU page-table dirty bits, so an explicit msync is the reliable writeback
// barrier on disk-backed filesystems (on tmpfs it is a near-free no-op) —
// matching write_file_atomic's fsync-before-rename durability.
void file_mmap_sync(const std::string& path) {
  std::lock_guard<std::mutex> lk(g_mmap_mu);
  auto it = g_mmap_cache.find(path);
  if (it != g_mmap_cache.end() && it->second.ptr) {
    if (msync(it->second.ptr, it->second.n, MS_SYNC) != 0)
      throw std::runtime_error("msync failed: " + path + ": " +
                               strerror(errno));
  }
  int fd = ::open(path.c_str(), O_RDONLY);
  if (fd >= 0) {
    (void)fsync(fd);
    ::close(fd);
  }
}


void gpu_memcpy_d2h(uintptr_t dst, uintptr_t src, i64 n, uintptr_t stream) {
  HIP_CHECK(hipMemcpyAsync((void*)dst, (const void*)src, (size_t)n,
                           hipMemcpyDeviceToHost, (hipStream_t)stream));
}

void gpu_memcpy_h2d(uintptr_t dst, uintptr_t src, i64 n, uintptr_t stream) {
  HIP_CHECK(hipMemcpyAsync((void*)dst, (const void*)src, (size_t)n,
                           hipMemcpyHostToDevice, (hipStream_t)stream));
}

}  // namespace

void register_gpu(py::module_& m) {
  m.def("gpu_device_count", []() {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    return (e == hipSuccess) ? n : -static_cast<int>(e);
  });
  m.def("gpu_crc_verify", &gpu_crc_verify, py::arg("data"), py::arg("off"),
        py::arg("len"), py::arg("R"), py::arg("err_out"), py::arg("stream"),
        py::arg("avg_bytes") = 0);
  m.def("gpu_scan_records", &gpu_scan_records, py::arg("data"), py::arg("off"),
        py::arg("len"), py::arg("R"), py::arg("fmt"), py::arg("schema_blob"),
        py::arg("F"), py::arg("stats"), py::arg("err"), py::arg("crc_err"),
        py::arg("stream"), py::arg("avg_bytes") = 0);
  m.def("gpu_extract_fields", &gpu_extract_fields, py::arg("data"),
        py::arg("R"), py::arg("F"), py::arg("stats"), py::arg("metas"),
        py::arg("meta_dev"), py::arg("err"), py::arg("stream"),
        py::arg("avg_bytes") = 0);
  m.def("gpu_size_records", &gpu_size_records);
  m.def("gpu_emit_records", &gpu_emit_records, py::arg("cols_dev"),
        py::arg("schema_blob"), py::arg("fmt"), py::arg("r0"), py::arg("R"),
        py::arg("frame_off"), py::arg("file"), py::arg("err"),
        py::arg("stream"), py::arg("avg_bytes") = 0);
  m.def("gpu_frame_bytes", &gpu_frame_bytes, py::arg("src"),
        py::arg("elem_off"), py::arg("frame_off"), py::arg("R"),
        py::arg("file"), py::arg("stream"), py::arg("avg_bytes") = 0);
  m.def("gpu_frame_scan_blocks", &gpu_frame_scan_blocks);
  m.def("gpu_frame_scan_count", &gpu_frame_scan_count);
  m.def("gpu_frame_scan_emit", &gpu_frame_scan_emit);
  m.def("gpu_gather_payloads", &gpu_gather_payloads, py::arg("data"),
        py::arg("off"), py::arg("len"), py::arg("dst_off"), py::arg("R"),
        py::arg("out"), py::arg("stream"), py::arg("avg_bytes") = 0);
  m.def("gpu_infer_codes", &gpu_infer_codes);
  m.def("gpu_scan_temp_bytes", &gpu_scan_temp_bytes);
  m.def("gpu_stat_scan_temp_bytes", &gpu_stat_scan_temp_bytes);
  m.def("gpu_stat_scans", &gpu_stat_scans);
  m.def("gpu_excl_sum_strided", &gpu_excl_sum_strided);
  m.def("file_mmap_pinned", &file_mmap_pinned, py::arg("path"), py::arg("n"),
        py::arg("writable"));
  m.def("file_mmap_drop", &file_mmap_drop);
  m.def("file_mmap_sync", &file_mmap_sync);
  m.def("gpu_memcpy_d2h", &gpu_memcpy_d2h);
  m.def("gpu_memcpy_h2d", &gpu_memcpy_h2d);
  m.def("gpu_devcols_bytes", &gpu_devcols_bytes);
  m.def("gpu_devmeta_bytes", []() { return sizeof(DevFieldDst); });
  m.def("gpu_fieldstat_words", []() { return sizeof(FieldStat) / 8; });
}
