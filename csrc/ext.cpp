// Python bindings + host-side (CPU) orchestration for the TFRecord codec.
//
// The CPU path here serves three roles:
//  - the no-GPU plumbing configuration (BASELINE.json config 1),
//  - the golden reference the HIP kernels are tested against,
//  - small-file metadata work (frame header scan) that stays on host even in
//    the GPU pipeline.
//
// GPU entry points live in csrc/hip/kernels.hip and are registered by
// register_gpu() when compiled in.

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <fcntl.h>
#include <unistd.h>

#include <atomic>
#include <condition_variable>
#include <cstring>
#include <functional>
#include <map>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

#include "codec_core.h"
#include "inflate_core.h"

namespace py = pybind11;
using namespace tfrec;

namespace {

// ---------------------------------------------------------------------------
// Parallel file IO: persistent pread/pwrite worker pool. A single-threaded
// write() to tmpfs tops out at ~9 GB/s (one kernel memcpy); splitting each
// chunk across workers scales that with cores. Used by the GPU engine to keep
// file IO off the critical path of the H2D/D2H pipeline
// (spark_tfrecord_amd/engine/gpu.py). Replaces what the reference delegates
// to Hadoop FS streams (DefaultSource.scala:118-136).
// ---------------------------------------------------------------------------

class IOPool {
 public:
  static IOPool& instance() {
    // leaked on purpose: workers block on the cv forever; destructing the
    // mutex/cv under them at exit would be UB
    static IOPool* pool = new IOPool(8);
    return *pool;
  }

  // op: false=read, true=write. Splits [off, off+n) across the workers and
  // blocks until all spans complete. Returns 0 or -errno.
  int run(bool is_write, int fd, u8* ptr, i64 n, i64 off) {
    return run_spans(n, [=](i64 o, i64 m) {
      return is_write ? do_write(fd, ptr + o, m, off + o)
                      : do_read(fd, ptr + o, m, off + o);
    });
  }

  template <typename Fn>
  int run_spans(i64 n, Fn&& fn) {
    return run_parts(n, 64 << 10, std::forward<Fn>(fn));
  }

  // Splits [0, n) into per-worker ranges of at least min_span and runs
  // fn(offset, count) on the pool; fn returns 0 or an error code (first
  // nonzero wins). Used for byte IO spans AND record ranges.
  template <typename Fn>
  int run_parts(i64 n, i64 min_span, Fn&& fn) {
    if (n <= 0) return 0;
    int nspan = (int)nthreads_;
    i64 span = (n + nspan - 1) / nspan;
    if (span < min_span) {
      nspan = (int)((n + min_span - 1) / min_span);
      if (nspan < 1) nspan = 1;
      span = (n + nspan - 1) / nspan;
    }
    if (nspan == 1) return fn((i64)0, n);  // run inline, skip the pool
    if (getenv("TFREC_POOL_DEBUG"))
      fprintf(stderr, "[pool] n=%lld nspan=%d span=%lld\n", (long long)n,
              nspan, (long long)span);
    std::atomic<int> err{0};
    std::atomic<int> left{nspan};
    {
      std::unique_lock<std::mutex> lk(mu_);
      for (int t = 0; t < nspan; ++t) {
        i64 o = (i64)t * span;
        i64 m = std::min(span, n - o);
        tasks_.push_back([=, &err, &left] {
          int e = fn(o, m);
          if (e) err.store(e, std::memory_order_relaxed);
          if (left.fetch_sub(1) == 1) {
            std::lock_guard<std::mutex> dl(done_mu_);
            done_cv_.notify_all();
          }
        });
      }
      cv_.notify_all();
    }
    std::unique_lock<std::mutex> dl(done_mu_);
    done_cv_.wait(dl, [&] { return left.load() == 0; });
    return err.load();
  }

 private:
  explicit IOPool(int n) : nthreads_(n) {
    for (int i = 0; i < n; ++i)
      workers_.emplace_back([this] { worker(); });
  }

  static int do_write(int fd, const u8* p, i64 n, i64 off) {
    while (n > 0) {
      ssize_t w = ::pwrite(fd, p, (size_t)n, (off_t)off);
      if (w < 0) {
        if (errno == EINTR) continue;
        return -errno;
      }
      p += w;
      n -= w;
      off += w;
    }
    return 0;
  }

  static int do_read(int fd, u8* p, i64 n, i64 off) {
    while (n > 0) {
      ssize_t r = ::pread(fd, p, (size_t)n, (off_t)off);
      if (r < 0) {
        if (errno == EINTR) continue;
        return -errno;
      }
      if (r == 0) return -EIO;  // unexpected EOF
      p += r;
      n -= r;
      off += r;
    }
    return 0;
  }

  void worker() {
    for (;;) {
      std::function<void()> task;
      {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [this] { return !tasks_.empty(); });
        task = std::move(tasks_.back());
        tasks_.pop_back();
      }
      task();
    }
  }

  size_t nthreads_;
  std::vector<std::thread> workers_;
  std::vector<std::function<void()>> tasks_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::mutex done_mu_;
  std::condition_variable done_cv_;
};

void pwrite_parallel(int fd, uintptr_t ptr, i64 n, i64 file_off) {
  int e;
  {
    py::gil_scoped_release rel;
    e = IOPool::instance().run(true, fd, (u8*)ptr, n, file_off);
  }
  if (e) throw std::runtime_error("pwrite failed: " + std::string(strerror(-e)));
}

void pread_parallel(int fd, uintptr_t ptr, i64 n, i64 file_off) {
  int e;
  {
    py::gil_scoped_release rel;
    e = IOPool::instance().run(false, fd, (u8*)ptr, n, file_off);
  }
  if (e) throw std::runtime_error("pread failed: " + std::string(strerror(-e)));
}

struct BufView {
  const u8* data;
  i64 size;
};

BufView as_bytes(const py::buffer& b, py::buffer_info& info) {
  info = b.request();
  if (info.itemsize != 1) throw std::invalid_argument("expected a byte buffer");
  return BufView{static_cast<const u8*>(info.ptr), static_cast<i64>(info.size)};
}

// ---------------------------------------------------------------------------
// Frame scan (host): sequential walk of [len][crc][payload][crc] frames.
// Mirrors the read-side behavior of the reference's record reader
// (TFRecordFileReader.scala:51 via tensorflow-hadoop). The headers are ~16
// bytes per record; this stays on host even for the GPU path, feeding the
// record-offset array that the kernels parallelize over (SURVEY.md §7
// "hard parts": TFRecord has no sync markers).
// ---------------------------------------------------------------------------

std::pair<py::array_t<i64>, py::array_t<i64>> scan_frames(py::buffer data,
                                                          bool verify_crc) {
  py::buffer_info info;
  BufView buf = as_bytes(data, info);
  std::vector<i64> offs, lens;
  i64 p = 0;
  while (p < buf.size) {
    if (buf.size - p < 12)
      throw std::runtime_error("corrupt TFRecord: truncated frame header at offset " +
                               std::to_string(p));
    u64 len;
    std::memcpy(&len, buf.data + p, 8);
    u32 len_crc;
    std::memcpy(&len_crc, buf.data + p + 8, 4);
    if (verify_crc && mask_crc(crc32c(buf.data + p, 8)) != len_crc)
      throw std::runtime_error("corrupt TFRecord: bad length CRC at offset " +
                               std::to_string(p));
    if (static_cast<u64>(buf.size - p - 12) < len + 4)
      throw std::runtime_error("corrupt TFRecord: truncated payload at offset " +
                               std::to_string(p));
    if (verify_crc) {
      u32 data_crc;
      std::memcpy(&data_crc, buf.data + p + 12 + len, 4);
      if (mask_crc(crc32c(buf.data + p + 12, len)) != data_crc)
        throw std::runtime_error("corrupt TFRecord: bad data CRC at offset " +
                                 std::to_string(p));
    }
    offs.push_back(p + 12);
    lens.push_back(static_cast<i64>(len));
    p += 12 + static_cast<i64>(len) + 4;
  }
  auto off_arr = py::array_t<i64>(offs.size());
  auto len_arr = py::array_t<i64>(lens.size());
  std::memcpy(off_arr.mutable_data(), offs.data(), offs.size() * 8);
  std::memcpy(len_arr.mutable_data(), lens.data(), lens.size() * 8);
  return {off_arr, len_arr};
}

// Fast header-only scan used by the GPU read path: offsets out, CRC checking
// deferred to the device kernel.
std::pair<py::array_t<i64>, py::array_t<i64>> scan_frame_headers(py::buffer data) {
  return scan_frames(data, false);
}

// ---------------------------------------------------------------------------
// Decode (host)
// ---------------------------------------------------------------------------

const char* err_name(int32_t e) {
  switch (e) {
    case ERR_TRUNCATED: return "truncated message";
    case ERR_BAD_VARINT: return "malformed varint";
    case ERR_KIND_MISMATCH: return "feature kind does not match schema";
    case ERR_BAD_WIRETYPE: return "unexpected wire type";
    default: return "codec error";
  }
}

py::list decode_records(py::buffer data, py::array_t<i64> rec_off,
                        py::array_t<i64> rec_len, py::bytes schema_blob,
                        int32_t fmt) {
  py::buffer_info info;
  BufView buf = as_bytes(data, info);
  std::string blob = schema_blob;
  SchemaView schema = schema_view(reinterpret_cast<const u8*>(blob.data()));
  const i64 R = rec_off.size();
  const int F = schema.nfields;
  auto off = rec_off.unchecked<1>();
  auto len = rec_len.unchecked<1>();

  // Pass A: structure scan -> per (record, field) stats. Pure C++ on raw
  // pointers: the GIL is released so multi-file readers can decode on
  // worker threads in parallel.
  std::vector<FieldStat> stats(static_cast<size_t>(R) * F);
  {
    std::atomic<i64> bad_rec{-1};
    std::atomic<int32_t> bad_rc{0};
    std::atomic<int> bad_field{-1};
    {
      py::gil_scoped_release rel;
      IOPool::instance().run_parts(R, 4096, [&](i64 o, i64 m) {
        for (i64 r = o; r < o + m; ++r) {
          FieldStat* st = stats.data() + r * F;
          for (int f = 0; f < F; ++f) field_stat_clear(&st[f]);
          int32_t rc = scan_record(buf.data, off(r), len(r), fmt, schema, st);
          for (int f = 0; rc == ERR_OK && f < F; ++f)
            if (st[f].err != ERR_OK) {
              rc = st[f].err;
              bad_field.store(f);
            }
          if (rc != ERR_OK) {
            bad_rec.store(r);
            bad_rc.store(rc);
            return 1;
          }
        }
        return 0;
      });
    }
    if (bad_rec.load() >= 0) {
      i64 r = bad_rec.load();
      int32_t rc = bad_rc.load();
      if (rc == ERR_KIND_MISMATCH && bad_field.load() >= 0) {
        int f = bad_field.load();
        std::string nm(reinterpret_cast<const char*>(schema.name(f)),
                       schema.name_len(f));
        throw std::runtime_error("Feature '" + nm +
                                 "' kind does not match requested data type (record " +
                                 std::to_string(r) + ")");
      }
      throw std::runtime_error(std::string("TFRecord decode failed in record ") +
                               std::to_string(r) + ": " + err_name(rc));
    }
  }

  // Prefix sums per field + allocation + pass B.
  py::list out;
  for (int f = 0; f < F; ++f) {
    const FieldDescRaw& fd = schema.fields[f];
    i64 total_vals = 0, total_bytes = 0, total_lists = 0;
    auto presence = py::array_t<u8>(R);
    auto row_off_arr = py::array_t<i64>(R + 1);
    u8* pres = presence.mutable_data();
    i64* row_off_p = row_off_arr.mutable_data();
    row_off_p[0] = 0;
    py::array_t<i64> list_off_arr(fd.is_seq ? R + 1 : 1);
    i64* list_off_p = list_off_arr.mutable_data();
    list_off_p[0] = 0;
    for (i64 r = 0; r < R; ++r) {
      const FieldStat& st = stats[r * F + f];
      pres[r] = st.pos >= 0 ? 1 : 0;
      total_vals += st.nvals;
      total_bytes += st.nbytes;
      total_lists += st.nlists;
      row_off_p[r + 1] = total_vals;
      if (fd.is_seq) list_off_p[r + 1] = total_lists;
    }

    DecodeDst dst{};
    py::array_t<i64> i64_vals(0);
    py::array_t<float> f32_vals(0);
    py::array_t<u8> bytes_data(0);
    py::array_t<i64> elem_len(0);
    py::array_t<i64> sub_count(0);
    if (fd.kind == KIND_INT64) {
      i64_vals = py::array_t<i64>(total_vals);
      dst.i64_vals = i64_vals.mutable_data();
    } else if (fd.kind == KIND_FLOAT) {
      f32_vals = py::array_t<float>(total_vals);
      dst.f32_vals = f32_vals.mutable_data();
    } else {
      bytes_data = py::array_t<u8>(total_bytes);
      elem_len = py::array_t<i64>(total_vals);
      dst.bytes_data = bytes_data.mutable_data();
      dst.elem_len = elem_len.mutable_data();
    }
    if (fd.is_seq) {
      sub_count = py::array_t<i64>(total_lists);
      dst.sub_count = sub_count.mutable_data();
    }

    // Extraction is record-independent once the destination bases are known:
    // prefix the byte/list bases (row_off already carries the value bases),
    // release the GIL and fan the record ranges across the worker pool.
    {
      std::vector<i64> byte_base_v(R);
      std::vector<i64> list_base_v(fd.is_seq ? R : 0);
      i64 bb = 0, lb = 0;
      for (i64 r = 0; r < R; ++r) {
        const FieldStat& st = stats[r * F + f];
        byte_base_v[r] = bb;
        bb += st.nbytes;
        if (fd.is_seq) {
          list_base_v[r] = lb;
          lb += st.nlists;
        }
      }
      const FieldStat* stp = stats.data();
      const u8* datap = buf.data;
      int rc;
      {
        py::gil_scoped_release rel;
        rc = IOPool::instance().run_parts(R, 4096, [&](i64 o, i64 m) {
          for (i64 r = o; r < o + m; ++r) {
            const FieldStat& st = stp[r * F + f];
            int32_t e = extract_field(datap, st.pos, st.len, fd.kind,
                                      fd.is_seq, dst, row_off_p[r],
                                      byte_base_v[r],
                                      fd.is_seq ? list_base_v[r] : 0);
            if (e != ERR_OK) return (int)-e;
          }
          return 0;
        });
      }
      if (rc != 0)
        throw std::runtime_error(std::string("TFRecord decode failed: ") +
                                 err_name((int32_t)-rc));
    }

    py::dict d;
    d["presence"] = presence;
    d["row_off"] = row_off_arr;
    if (fd.kind == KIND_INT64) d["values"] = i64_vals;
    else if (fd.kind == KIND_FLOAT) d["values"] = f32_vals;
    else {
      d["values"] = bytes_data;
      d["elem_len"] = elem_len;  // Python cumsums to elem_off
    }
    if (fd.is_seq) {
      d["list_off"] = list_off_arr;
      d["sub_count"] = sub_count;  // Python cumsums to sub_off
    }
    out.append(d);
  }
  return out;
}

// ---------------------------------------------------------------------------
// Encode (host): columnar wire-form -> full framed file image.
// ---------------------------------------------------------------------------

template <typename T>
const T* opt_ptr(const py::dict& d, const char* key) {
  if (!d.contains(key)) return nullptr;
  auto arr = d[key].cast<py::array_t<T>>();
  return arr.data();
}

// Keeps the cast py::array objects alive while we hold raw pointers.
struct ColKeepAlive {
  std::vector<py::object> refs;
  template <typename T>
  const T* get(const py::dict& d, const char* key) {
    if (!d.contains(key)) return nullptr;
    py::array_t<T, py::array::c_style | py::array::forcecast> arr =
        py::cast<py::array_t<T, py::array::c_style | py::array::forcecast>>(d[key]);
    refs.push_back(arr);
    return arr.data();
  }
};

py::bytes encode_records(py::bytes schema_blob, int32_t fmt, py::list col_dicts,
                         i64 R) {
  std::string blob = schema_blob;
  SchemaView schema = schema_view(reinterpret_cast<const u8*>(blob.data()));
  const int F = schema.nfields;
  if (static_cast<int>(col_dicts.size()) != F)
    throw std::invalid_argument("column count does not match schema");

  ColKeepAlive keep;
  std::vector<FieldColumn> cols(F);
  for (int f = 0; f < F; ++f) {
    py::dict d = col_dicts[f].cast<py::dict>();
    FieldColumn& c = cols[f];
    c.kind = schema.fields[f].kind;
    c.is_seq = schema.fields[f].is_seq;
    c.presence = keep.get<u8>(d, "presence");
    c.row_off = keep.get<i64>(d, "row_off");
    c.list_off = keep.get<i64>(d, "list_off");
    c.sub_off = keep.get<i64>(d, "sub_off");
    c.elem_off = keep.get<i64>(d, "elem_off");
    c.bytes_data = keep.get<u8>(d, "values_bytes");
    c.i64_vals = keep.get<i64>(d, "values_i64");
    c.f32_vals = keep.get<float>(d, "values_f32");
    if (!c.presence || !c.row_off)
      throw std::invalid_argument("column missing presence/row_off");
  }

  // Both passes are pure C++ over raw pointers: the GIL is released and the
  // records parallelize across the worker pool (sizes and emits are
  // record-independent; only the frame-offset prefix sum is sequential).
  std::string out;
  bool emit_mismatch = false;
  {
    py::gil_scoped_release rel;
    std::vector<i64> frame_off(R + 1);
    std::vector<i64> psize(R);
    const FieldColumn* cp = cols.data();
    IOPool::instance().run_parts(R, 4096, [&](i64 o, i64 m) {
      for (i64 r = o; r < o + m; ++r)
        psize[r] = record_payload_size(cp, schema, fmt, r);
      return 0;
    });
    frame_off[0] = 0;
    for (i64 r = 0; r < R; ++r)
      frame_off[r + 1] = frame_off[r] + psize[r] + kFrameOverhead;
    i64 total = frame_off[R];

    // Pass B: fused emit (each output window stored once, CRC from the same
    // registers — codec_core.h WriteCur; identical bytes to the plain form).
    out.resize(static_cast<size_t>(total), '\0');
    u8* file = reinterpret_cast<u8*>(out.data());
    int rc = IOPool::instance().run_parts(R, 4096, [&](i64 o, i64 m) {
      for (i64 r = o; r < o + m; ++r) {
        u8* dst = file + frame_off[r] + 12;
        u32 crc = 0;
        i64 emitted = emit_record_payload_fused(dst, cp, schema, fmt, r, &crc,
                                                kCrcTables.t);
        if (emitted != psize[r]) return 1;
        write_frame_header_footer_crc(file, frame_off[r], psize[r], crc,
                                      kCrcTables.t);
      }
      return 0;
    });
    emit_mismatch = rc != 0;
  }
  if (emit_mismatch)
    throw std::runtime_error("internal error: emit size mismatch");
  return py::bytes(out);
}

// ByteArray write path: frame raw binary payloads directly
// (reference: TFRecordSerializer.scala:16-18 + TFRecordOutputWriter.scala:28-29).
py::bytes frame_byte_arrays(py::buffer data, py::array_t<i64> elem_off) {
  py::buffer_info info;
  BufView buf = as_bytes(data, info);
  auto off = elem_off.unchecked<1>();
  i64 R = elem_off.size() - 1;
  i64 total = 0;
  for (i64 r = 0; r < R; ++r) total += (off(r + 1) - off(r)) + kFrameOverhead;
  std::string out(static_cast<size_t>(total), '\0');
  u8* file = reinterpret_cast<u8*>(out.data());
  i64 fo = 0;
  for (i64 r = 0; r < R; ++r) {
    i64 n = off(r + 1) - off(r);
    std::memcpy(file + fo + 12, buf.data + off(r), static_cast<size_t>(n));
    write_frame_header_footer(file, fo, n, kCrcTables.t);
    fo += n + kFrameOverhead;
  }
  return py::bytes(out);
}

// ---------------------------------------------------------------------------
// Schema inference (host): per-record lattice codes merged with max.
// Mirrors TensorFlowInferSchema.scala:75-118 + :132-188:
//   per row: n==0 -> null, n==1 -> scalar, n>1 -> array; FeatureLists always
//   infer 2-D; bytes infer as String; merge = lattice max.
// Returns {name: code} with codes from schema.py lattice_code (seq fields
// offset by their 2-D position in the lattice).
// ---------------------------------------------------------------------------

// kind (1=bytes,2=float,3=int64) -> rank 3/2/1; code = rank, +3 if multi, +6 if seq
inline int lattice_code_for(int32_t kind, bool multi, bool seq) {
  int rank = 4 - kind;  // int64->1 float->2 bytes->3
  if (seq) return 6 + rank;
  if (multi) return 3 + rank;
  return rank;
}

void infer_features_body(const u8* p, const u8* end, bool seq,
                         std::map<std::string, int>& codes) {
  while (p < end) {
    u64 tag;
    p = read_varint(p, end, &tag);
    if (!p) throw std::runtime_error("malformed record during schema inference");
    u32 fieldno = static_cast<u32>(tag >> 3);
    u32 wt = static_cast<u32>(tag & 7);
    if (fieldno != 1 || wt != 2) {
      p = skip_field(p, end, wt);
      if (!p) throw std::runtime_error("malformed record during schema inference");
      continue;
    }
    u64 entry_len;
    p = read_varint(p, end, &entry_len);
    if (!p || static_cast<u64>(end - p) < entry_len)
      throw std::runtime_error("malformed record during schema inference");
    const u8* ep = p;
    const u8* ee = p + entry_len;
    p = ee;
    std::string key;
    const u8* val = nullptr;
    u64 val_len = 0;
    while (ep < ee) {
      u64 etag;
      ep = read_varint(ep, ee, &etag);
      if (!ep) throw std::runtime_error("malformed map entry during schema inference");
      u32 efn = static_cast<u32>(etag >> 3);
      u32 ewt = static_cast<u32>(etag & 7);
      if (efn == 1 && ewt == 2) {
        u64 klen;
        ep = read_varint(ep, ee, &klen);
        if (!ep || static_cast<u64>(ee - ep) < klen)
          throw std::runtime_error("malformed map entry during schema inference");
        key.assign(reinterpret_cast<const char*>(ep), klen);
        ep += klen;
      } else if (efn == 2 && ewt == 2) {
        ep = read_varint(ep, ee, &val_len);
        if (!ep || static_cast<u64>(ee - ep) < val_len)
          throw std::runtime_error("malformed map entry during schema inference");
        val = ep;
        ep += val_len;
      } else {
        ep = skip_field(ep, ee, ewt);
        if (!ep) throw std::runtime_error("malformed map entry during schema inference");
      }
    }
    if (key.empty() && !val) continue;
    int code = 0;
    if (val) {
      if (!seq) {
        int32_t kf = 0;
        i64 nvals = 0, nbytes = 0;
        int32_t rc = scan_feature_body(val, val + val_len, -1, &kf, &nvals, &nbytes);
        if (rc != ERR_OK)
          throw std::runtime_error("malformed Feature during schema inference");
        if (kf != 0 && nvals > 0) code = lattice_code_for(kf, nvals > 1, false);
        // n==0 or no kind set -> null (code 0): TensorFlowInferSchema.scala:147-188
      } else {
        // FeatureList: any content infers as Array(Array(T)) regardless of
        // lengths (TensorFlowInferSchema.scala:98-118).
        const u8* lp = val;
        const u8* le = val + val_len;
        int32_t kind_seen = 0;
        bool any_vals = false;
        while (lp < le) {
          u64 ltag;
          lp = read_varint(lp, le, &ltag);
          if (!lp) throw std::runtime_error("malformed FeatureList during inference");
          u32 lfn = static_cast<u32>(ltag >> 3);
          u32 lwt = static_cast<u32>(ltag & 7);
          if (lfn == 1 && lwt == 2) {
            u64 flen;
            lp = read_varint(lp, le, &flen);
            if (!lp || static_cast<u64>(le - lp) < flen)
              throw std::runtime_error("malformed FeatureList during inference");
            int32_t kf = 0;
            i64 nvals = 0, nbytes = 0;
            int32_t rc = scan_feature_body(lp, lp + flen, -1, &kf, &nvals, &nbytes);
            if (rc != ERR_OK)
              throw std::runtime_error("malformed Feature during schema inference");
            if (kf) {
              kind_seen = std::max(kind_seen, lattice_code_for(kf, false, false));
              if (nvals >= 0) any_vals = true;
            }
            lp += flen;
          } else {
            lp = skip_field(lp, le, lwt);
            if (!lp) throw std::runtime_error("malformed FeatureList during inference");
          }
        }
        (void)any_vals;
        if (kind_seen) code = 6 + kind_seen;  // rank embedded in kind_seen
      }
    }
    auto it = codes.find(key);
    if (it == codes.end()) codes[key] = code;
    else it->second = std::max(it->second, code);
  }
}

py::dict infer_schema_codes(py::buffer data, py::array_t<i64> rec_off,
                            py::array_t<i64> rec_len, int32_t fmt) {
  py::buffer_info info;
  BufView buf = as_bytes(data, info);
  auto off = rec_off.unchecked<1>();
  auto len = rec_len.unchecked<1>();
  std::map<std::string, int> ctx_codes, seq_codes;
  for (i64 r = 0; r < rec_off.size(); ++r) {
    const u8* p = buf.data + off(r);
    const u8* end = p + len(r);
    while (p < end) {
      u64 tag;
      p = read_varint(p, end, &tag);
      if (!p) throw std::runtime_error("malformed record during schema inference");
      u32 fieldno = static_cast<u32>(tag >> 3);
      u32 wt = static_cast<u32>(tag & 7);
      bool is_features = fieldno == 1;
      bool is_fl = (fmt == FMT_SEQUENCE && fieldno == 2);
      if ((is_features || is_fl) && wt == 2) {
        u64 blen;
        p = read_varint(p, end, &blen);
        if (!p || static_cast<u64>(end - p) < blen)
          throw std::runtime_error("malformed record during schema inference");
        infer_features_body(p, p + blen, is_fl, is_fl ? seq_codes : ctx_codes);
        p += blen;
      } else {
        p = skip_field(p, end, wt);
        if (!p) throw std::runtime_error("malformed record during schema inference");
      }
    }
  }
  // Feature names decode with replacement characters on invalid UTF-8,
  // like the JVM's new String(bytes, UTF_8) in the reference — corrupt
  // names surface as mangled fields (and CRC failures downstream), not
  // as a decode exception here.
  auto safe_str = [](const std::string& v) {
    return py::reinterpret_steal<py::str>(
        PyUnicode_DecodeUTF8(v.data(), (Py_ssize_t)v.size(), "replace"));
  };
  py::dict out;
  for (auto& kv : ctx_codes) out[safe_str(kv.first)] = kv.second;
  for (auto& kv : seq_codes) {
    // A name can only be context or sequence within one record type read.
    out[safe_str(kv.first)] = kv.second;
  }
  return out;
}

// TEST-ONLY: run pass-A on host in either form and return the raw stats +
// per-record payload CRCs, so the fused cursor scan can be differentially
// tested against the two-pass reference on arbitrary inputs.
py::dict scan_stats_debug(py::buffer data, py::array_t<i64> rec_off,
                          py::array_t<i64> rec_len, int32_t fmt,
                          py::bytes schema_blob, bool fused) {
  py::buffer_info info;
  BufView buf = as_bytes(data, info);
  std::string blob = schema_blob;
  SchemaView schema = schema_view(reinterpret_cast<const u8*>(blob.data()));
  const int F = schema.nfields;
  auto off = rec_off.unchecked<1>();
  auto len = rec_len.unchecked<1>();
  i64 R = rec_off.size();
  auto stats_arr = py::array_t<i64>({R, (i64)F, (i64)6});
  auto crc_arr = py::array_t<i64>(R);
  auto rc_arr = py::array_t<i64>(R);
  auto* st_all = reinterpret_cast<FieldStat*>(stats_arr.mutable_data());
  for (i64 r = 0; r < R; ++r) {
    FieldStat* st = st_all + r * F;
    for (int f = 0; f < F; ++f) field_stat_clear(&st[f]);
    int32_t rc;
    u32 crc = 0;
    if (fused) {
      rc = scan_record_fused(buf.data, off(r), len(r), fmt, schema, st, &crc,
                             kCrcTables.t);
      if (rc == ERR_RETRY_UNFUSED) {
        for (int f = 0; f < F; ++f) field_stat_clear(&st[f]);
        rc = scan_record(buf.data, off(r), len(r), fmt, schema, st);
        crc = crc32c(buf.data + off(r), (size_t)len(r));
      }
    } else {
      rc = scan_record(buf.data, off(r), len(r), fmt, schema, st);
      crc = crc32c(buf.data + off(r), (size_t)len(r));
    }
    crc_arr.mutable_at(r) = rc == ERR_OK ? (i64)crc : -1;
    rc_arr.mutable_at(r) = rc;
  }
  py::dict out;
  out["stats"] = stats_arr;
  out["crc"] = crc_arr;
  out["rc"] = rc_arr;
  return out;
}

// Host run of the DEFLATE segment inflater core shared with the gfx950
// kernel (csrc/inflate_core.h): backs CPU tests of the device algorithm.
py::bytes host_inflate_segment(py::buffer comp, i64 expect) {
  auto info = comp.request();
  std::string out(static_cast<size_t>(expect), '\0');
  tfrec::inflate::LaneScratch L;
  uint16_t lit_tab[tfrec::inflate::kLitTabSize];
  uint16_t dist_tab[tfrec::inflate::kDistTabSize];
  int rc = tfrec::inflate::inflate_one(
      static_cast<const u8*>(info.ptr), static_cast<i64>(info.size),
      reinterpret_cast<u8*>(&out[0]), expect, L, lit_tab, dist_tab);
  if (rc)
    throw std::runtime_error("inflate_one failed: cause " + std::to_string(rc));
  return py::bytes(out);
}

u32 crc32c_py(py::buffer data) {
  py::buffer_info info;
  BufView buf = as_bytes(data, info);
  return crc32c(buf.data, static_cast<size_t>(buf.size));
}

u32 masked_crc32c_py(py::buffer data) {
  py::buffer_info info;
  BufView buf = as_bytes(data, info);
  return masked_crc32c(buf.data, static_cast<size_t>(buf.size));
}

}  // namespace

void register_gpu(py::module_& m);      // defined in csrc/hip/kernels.hip
void register_inflate(py::module_& m);  // defined in csrc/hip/inflate.hip

PYBIND11_MODULE(_native, m) {
  m.doc() = "MI355X-native TFRecord codec (host + gfx950 kernels)";
  m.def("pwrite_parallel", &pwrite_parallel, py::arg("fd"), py::arg("ptr"),
        py::arg("n"), py::arg("file_off"),
        "Multi-threaded pwrite of [ptr, ptr+n) to fd at file_off");
  m.def("pread_parallel", &pread_parallel, py::arg("fd"), py::arg("ptr"),
        py::arg("n"), py::arg("file_off"),
        "Multi-threaded pread of n bytes at file_off into ptr");
  m.def("scan_stats_debug", &scan_stats_debug, py::arg("data"),
        py::arg("rec_off"), py::arg("rec_len"), py::arg("fmt"),
        py::arg("schema_blob"), py::arg("fused"),
        "TEST-ONLY: pass-A stats + payload CRCs (fused cursor vs two-pass)");
  m.def("crc32c", &crc32c_py, "CRC32C (Castagnoli) of a byte buffer");
  m.def("masked_crc32c", &masked_crc32c_py, "TFRecord-masked CRC32C");
  m.def("crc32c_combine",
        [](u32 c1, u32 c2, u64 len2) { return crc32c_combine(c1, c2, len2); },
        "crc(A||B) from crc(A), crc(B), len(B) (GF(2) shift operator)");
  m.def("host_inflate_segment", &host_inflate_segment, py::arg("comp"),
        py::arg("expect"),
        "TEST-ONLY: run the device inflater's core on host for one segment");
  m.def("crc32c_combine_fast",
        [](u32 c1, u32 c2, u64 len2) { return crc32c_combine_fast(c1, c2, len2); },
        "crc(A||B) via register-only GF(2^32) field multiply (wave-CRC form)");
  m.def("scan_frames", &scan_frames, py::arg("data"), py::arg("verify_crc") = true,
        "Scan TFRecord frames -> (payload offsets, payload lengths)");
  m.def("scan_frame_headers", &scan_frame_headers, py::arg("data"),
        "Header-only frame scan (CRC verification deferred to the GPU)");
  m.def("decode_records", &decode_records, py::arg("data"), py::arg("rec_off"),
        py::arg("rec_len"), py::arg("schema_blob"), py::arg("fmt"),
        "Decode Example/SequenceExample payloads into columnar wire-form");
  m.def("encode_records", &encode_records, py::arg("schema_blob"), py::arg("fmt"),
        py::arg("columns"), py::arg("num_rows"),
        "Encode columnar wire-form into a framed TFRecord file image");
  m.def("frame_byte_arrays", &frame_byte_arrays, py::arg("data"), py::arg("elem_off"),
        "Frame raw byte payloads (ByteArray record type)");
  m.def("infer_schema_codes", &infer_schema_codes, py::arg("data"), py::arg("rec_off"),
        py::arg("rec_len"), py::arg("fmt"),
        "Per-feature type-lattice codes for schema inference");
  m.attr("FMT_EXAMPLE") = static_cast<int>(FMT_EXAMPLE);
  m.attr("FMT_SEQUENCE") = static_cast<int>(FMT_SEQUENCE);
  m.attr("FMT_BYTE_ARRAY") = static_cast<int>(FMT_BYTE_ARRAY);
#ifdef TFREC_WITH_HIP
  register_gpu(m);
  register_inflate(m);
  m.attr("HAS_GPU_KERNELS") = true;
#else
  m.attr("HAS_GPU_KERNELS") = false;
#endif
}
