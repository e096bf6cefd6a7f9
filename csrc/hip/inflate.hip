// gfx950 DEFLATE inflater: one compressed segment per LANE.
//
// The engine writes gzip as a single member whose deflate stream carries
// Z_FULL_FLUSH sync points and an FEXTRA table of per-segment extents
// (spark_tfrecord_amd/io/paths.py compress_bytes). Each segment is an
// independent raw-deflate sub-stream (dictionary reset at the flush), so
// inflation parallelizes segment-per-lane with output offsets known up
// front from the table — one kernel launch inflates a whole batch of files.
//
// Huffman decode is inherently bit-serial, so parallelism comes purely from
// segment count (32 KiB segments => ~7000 waves per 215 MB batch; the
// reader batches many files into one launch). One segment per WAVE: the 64
// lanes run the decode in lockstep on the same bitstream (wave-uniform =>
// scalarized, no SIMT divergence) and split the bulk copies; the ~1 KB
// canonical-table scratch lives once per wave in LDS. See docs/KERNELS.md
// for the measured decomposition history (per-LANE segments were 5x
// slower from divergence alone).
//
// This is NOT a general gzip: host zlib remains the fallback for foreign
// files (no table), multi-member streams, or any kernel-reported error.
// Reference behavior being replaced: Hadoop CodecStreams gzip read,
// DefaultSource.scala:95-102 (codec by extension).

#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <cstdlib>

#include "../codec_core.h"
#include "../inflate_core.h"

namespace py = pybind11;
using namespace tfrec;

namespace {

#define HIPI_CHECK(expr)                                                       \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess)                                                      \
      throw std::runtime_error(std::string("HIP error: ") +                    \
                               hipGetErrorString(_e));                         \
  } while (0)

using tfrec::inflate::LaneScratch;
using tfrec::inflate::inflate_one;

// One segment per WAVE: all 64 lanes run the decode in lockstep on the SAME
// bitstream — every value is wave-uniform (scalarized by the compiler, no
// SIMT divergence; the per-LANE decomposition paid ~5x executing the union
// of 64 independent streams' branches), and the lanes split the bulk
// copies. 4 waves per block => 4 KiB LDS, so occupancy is wave-limited,
// not LDS-limited.
// NOTE: amdgpu_waves_per_eu(6) was tried (80 VGPRs, 6 waves/SIMD) and
// measured ~25% SLOWER — the forced VGPR cap introduced vector spills on
// the serial decode chain, which cost more than the extra wave hid.
//
// TWO segments per wave (one per 32-lane HALF). PMC showed the "wave-
// uniform" decode was never actually scalarized: all 64 lanes execute the
// same vector instructions redundantly on one stream. Giving each half its
// own stream makes those SAME instructions carry two streams — near-free
// except (a) EXEC-mask divergence where the halves take different paths
// (rare on literal-heavy data; amortized on headers), (b) per-half LDS
// scratch/tables (~28 KiB/block, still under the 5-block register cap),
// (c) half-width bulk copies for stored blocks/long matches. The launcher
// pairs adjacent segments, which the callers emit in like-sized order.
__global__ void __launch_bounds__(256) inflate_segments_kernel(
    const u8* __restrict__ comp, const i64* __restrict__ in_off,
    const i64* __restrict__ in_len, const i64* __restrict__ out_off,
    const i64* __restrict__ out_len, i64 nseg, u8* __restrict__ out,
    unsigned long long* __restrict__ err) {
  __shared__ LaneScratch S[8];
  __shared__ uint16_t T[8][tfrec::inflate::kLitTabSize];
  __shared__ uint16_t D[8][tfrec::inflate::kDistTabSize];
  const int half = threadIdx.x >> 5;       // 0..7 within the block
  const int lane = threadIdx.x & 31;
  i64 stream0 = blockIdx.x * 8 + half;
  i64 nstreams = (i64)gridDim.x * 8;
  for (i64 seg = stream0; seg < nseg; seg += nstreams) {
    int rc = inflate_one<32>(comp + in_off[seg], in_len[seg],
                             out + out_off[seg], out_len[seg], S[half],
                             T[half], D[half], lane);
    if (rc && lane == 0)
      atomicMin(err, ((unsigned long long)(seg + 1) << 8) | (u32)rc);
  }
}

// FOUR streams per wave (16 lanes each) with narrower tables (9-bit lit,
// 7-bit dist) so LDS still fits 4 blocks/CU at occupancy 4 — twice the
// streams in flight of the half-wave kernel. MEASURED: no gain on
// literal-heavy data (21.0 vs 21.7M rows/s on config 5) and 33% slower on
// match-heavy text — 4-way refill/flush divergence eats the extra latency
// hiding. Kept behind TFREC_INFLATE_STREAMS=4 for measurement; the auto
// router never picks it.
__global__ void __launch_bounds__(256) inflate_segments_kernel4(
    const u8* __restrict__ comp, const i64* __restrict__ in_off,
    const i64* __restrict__ in_len, const i64* __restrict__ out_off,
    const i64* __restrict__ out_len, i64 nseg, u8* __restrict__ out,
    unsigned long long* __restrict__ err) {
  __shared__ LaneScratch S[16];
  __shared__ uint16_t T[16][1 << 9];
  __shared__ uint16_t D[16][1 << 7];
  const int q = threadIdx.x >> 4;          // 0..15 within the block
  const int lane = threadIdx.x & 15;
  i64 stream0 = blockIdx.x * 16 + q;
  i64 nstreams = (i64)gridDim.x * 16;
  for (i64 seg = stream0; seg < nseg; seg += nstreams) {
    int rc = inflate_one<16, 9, 7>(comp + in_off[seg], in_len[seg],
                                   out + out_off[seg], out_len[seg], S[q],
                                   T[q], D[q], lane);
    if (rc && lane == 0)
      atomicMin(err, ((unsigned long long)(seg + 1) << 8) | (u32)rc);
  }
}

// Single-stream-per-wave variant kept for A/B measurement
// (TFREC_INFLATE_STREAMS=1).
__global__ void __launch_bounds__(256) inflate_segments_kernel1(
    const u8* __restrict__ comp, const i64* __restrict__ in_off,
    const i64* __restrict__ in_len, const i64* __restrict__ out_off,
    const i64* __restrict__ out_len, i64 nseg, u8* __restrict__ out,
    unsigned long long* __restrict__ err) {
  __shared__ LaneScratch S[4];
  __shared__ uint16_t T[4][tfrec::inflate::kLitTabSize];
  __shared__ uint16_t D[4][tfrec::inflate::kDistTabSize];
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  i64 wave = blockIdx.x * 4 + wid;
  i64 nwaves = (i64)gridDim.x * 4;
  for (i64 seg = wave; seg < nseg; seg += nwaves) {
    int rc = inflate_one(comp + in_off[seg], in_len[seg], out + out_off[seg],
                         out_len[seg], S[wid], T[wid], D[wid], lane);
    if (rc && lane == 0)
      atomicMin(err, ((unsigned long long)(seg + 1) << 8) | (u32)rc);
  }
}

// streams_per_wave: 1 = whole-wave (best for match-heavy segments — wide
// cooperative copies, zero divergence), 2 = half-wave pairs (best for
// literal-heavy segments — the same instructions carry two streams). The
// Python caller routes each segment by its compression ratio.
void gpu_inflate_segments(uintptr_t comp, uintptr_t in_off, uintptr_t in_len,
                          uintptr_t out_off, uintptr_t out_len, i64 nseg,
                          uintptr_t out, uintptr_t err, uintptr_t stream,
                          int streams_per_wave) {
  if (nseg <= 0) return;
  if (streams_per_wave >= 4) {
    i64 blocks = (nseg + 15) / 16;  // 16 quarter-wave streams per block
    if (blocks > 16384) blocks = 16384;
    hipLaunchKernelGGL(inflate_segments_kernel4, dim3((uint32_t)blocks),
                       dim3(256), 0, (hipStream_t)stream, (const u8*)comp,
                       (const i64*)in_off, (const i64*)in_len,
                       (const i64*)out_off, (const i64*)out_len, nseg,
                       (u8*)out, (unsigned long long*)err);
    HIPI_CHECK(hipGetLastError());
    return;
  }
  if (streams_per_wave <= 1) {
    i64 blocks = (nseg + 3) / 4;  // 4 waves (segments) per block
    if (blocks > 16384) blocks = 16384;
    hipLaunchKernelGGL(inflate_segments_kernel1, dim3((uint32_t)blocks),
                       dim3(256), 0, (hipStream_t)stream, (const u8*)comp,
                       (const i64*)in_off, (const i64*)in_len,
                       (const i64*)out_off, (const i64*)out_len, nseg,
                       (u8*)out, (unsigned long long*)err);
    HIPI_CHECK(hipGetLastError());
    return;
  }
  i64 blocks = (nseg + 7) / 8;  // 8 half-wave streams per block
  if (blocks > 16384) blocks = 16384;
  hipLaunchKernelGGL(inflate_segments_kernel, dim3((uint32_t)blocks), dim3(256),
                     0, (hipStream_t)stream, (const u8*)comp,
                     (const i64*)in_off, (const i64*)in_len,
                     (const i64*)out_off, (const i64*)out_len, nseg, (u8*)out,
                     (unsigned long long*)err);
  HIPI_CHECK(hipGetLastError());
}

}  // namespace

void register_inflate(py::module_& m) {
  m.def("gpu_inflate_segments", &gpu_inflate_segments, py::arg("comp"),
        py::arg("in_off"), py::arg("in_len"), py::arg("out_off"),
        py::arg("out_len"), py::arg("nseg"), py::arg("out"), py::arg("err"),
        py::arg("stream"), py::arg("streams_per_wave") = 1,
        "Inflate full-flush deflate segments (1 = one per wave, 2 = one per "
        "half-wave); err[0] (init ~0ull) collects ((seg+1)<<8)|cause of the "
        "first failure");
}
