// kernels_pq.hip — GPU Parquet page decompression + decode staging.
//
// The config-3 scan is bound by HOST snappy decompression (~90% of chunk
// decode time: tools/pq_prof). Pages decompress independently, so the wide
// work moves to the GPU: one WAVE per page runs the (inherently sequential)
// snappy token loop with uniform control flow — the tag/length computations
// are wave-uniform scalar loads, the literal/copy byte movement is
// lane-parallel — and thousands of pages run concurrently. The def-level
// section of each v1 page (RLE/bit-packed hybrid, parquet spec; host
// restatement parquet.cpp rle_bp_runs) is parsed by the same wave, emitting
// validity bits straight into the chunk's bitmap. A second pass computes
// per-page dense offsets (exclusive scan of non-null counts within each
// chunk), and a third compacts each page's dense value section into the
// chunk's dense values blob (the same dense-values + validity contract the
// host decode produces, so the engine's scatter path is unchanged).
//
// Scope guard (host pre-scan, parquet.cpp): SNAPPY codec, fixed-width
// types, v1 PLAIN data pages, <= PQ_MAX_PAGE_VALUES values per page.
#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#include "kernels.h"

namespace auron {

namespace {
inline void check_launch_pq(const char* name) {
  hipError_t e = hipGetLastError();
  if (e != hipSuccess)
    throw std::runtime_error(std::string(name) + ": " + hipGetErrorString(e));
}
}  // namespace

// ---- snappy block decode, one wave per page --------------------------------
// Tag walk is wave-uniform (every lane loads the same bytes — scalarized by
// the compiler); copies are lane-parallel. Returns false on malformed input.
__device__ static bool dev_snappy_page(const uint8_t* __restrict__ src,
                                       uint32_t n, uint8_t* __restrict__ dst,
                                       uint32_t out_len) {
  const int lane = (int)(threadIdx.x & 63);
  uint32_t pos = 0;
  {
    uint64_t hdr = 0;
    int shift = 0;
    bool got = false;
    while (pos < n) {
      uint8_t b = src[pos++];
      hdr |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) {
        got = true;
        break;
      }
      shift += 7;
    }
    if (!got || hdr != out_len) return false;
  }
  uint32_t op = 0;
  while (pos < n && op < out_len) {
    uint8_t tag = src[pos++];
    int type = tag & 3;
    if (type == 0) {  // literal
      uint32_t len = (tag >> 2) + 1;
      if (len > 60) {
        int nb = (int)len - 60;
        if (pos + nb > n) return false;
        len = 0;
        for (int i = 0; i < nb; i++) len |= (uint32_t)src[pos++] << (8 * i);
        len += 1;
      }
      if (pos + len > n || op + len > out_len) return false;
      for (uint32_t d = lane; d < len; d += 64) dst[op + d] = src[pos + d];
      pos += len;
      op += len;
    } else {
      uint32_t len, off;
      if (type == 1) {
        if (pos + 1 > n) return false;
        len = ((tag >> 2) & 7) + 4;
        off = ((uint32_t)(tag >> 5) << 8) | src[pos];
        pos += 1;
      } else if (type == 2) {
        if (pos + 2 > n) return false;
        len = (tag >> 2) + 1;
        off = (uint32_t)src[pos] | ((uint32_t)src[pos + 1] << 8);
        pos += 2;
      } else {
        if (pos + 4 > n) return false;
        len = (tag >> 2) + 1;
        off = 0;
        for (int i = 0; i < 4; i++) off |= (uint32_t)src[pos + i] << (8 * i);
        pos += 4;
      }
      if (off == 0 || off > op || op + len > out_len) return false;
      // pattern-doubling: each round copies min(avail, rest) bytes from the
      // already-valid region, so overlapping copies replicate correctly
      uint32_t start = op - off;
      uint32_t done = 0, avail = off;
      while (done < len) {
        uint32_t take = len - done < avail ? len - done : avail;
        for (uint32_t d = lane; d < take; d += 64)
          dst[op + done + d] = dst[start + d];
        // dst[start + take .. ) stays the same pattern: extending the window
        // keeps sources valid because take bytes were just appended
        done += take;
        avail += take;
        __builtin_amdgcn_wave_barrier();  // lanes agree before re-reading
      }
      op += len;
    }
  }
  return op == out_len && pos <= n;
}

// ---- def-level walk (bit width 1) ------------------------------------------
// Walks the RLE/bit-packed hybrid (host restatement: parquet.cpp
// rle_bp_runs) and ORs validity bits into the chunk bitmap at absolute bit
// positions (pages are not byte-aligned; boundary bytes are shared across
// pages, hence atomicOr everywhere — ~KBs per page, L2-resident).
// Returns the ones count, or sets *bad.
__device__ static uint32_t dev_def_walk(const uint8_t* __restrict__ p,
                                        uint32_t len, uint32_t count,
                                        uint8_t* __restrict__ valid,
                                        uint64_t abs_bit_base, bool* bad) {
  const int lane = (int)(threadIdx.x & 63);
  uint32_t pos = 0, filled = 0, ones = 0;
  while (filled < count) {
    if (pos >= len) {
      *bad = true;
      return 0;
    }
    uint64_t header = 0;
    int shift = 0;
    bool got = false;
    while (pos < len) {
      uint8_t b = p[pos++];
      header |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) {
        got = true;
        break;
      }
      shift += 7;
    }
    if (!got) {
      *bad = true;
      return 0;
    }
    uint64_t B = abs_bit_base + filled;  // absolute dest bit of this run
    if (header & 1) {                    // bit-packed group (bw == 1)
      uint32_t nbits = (uint32_t)(header >> 1) * 8;
      uint32_t bytes = nbits / 8;
      if (pos + bytes > len) {
        *bad = true;
        return 0;
      }
      uint32_t take = nbits < count - filled ? nbits : count - filled;
      // per-destination-byte gather from up to two source bytes
      uint64_t first = B >> 3, last = (B + take - 1) >> 3;
      int r = (int)(B & 7);  // src bit 0 lands at dest bit r
      uint32_t local_ones = 0;
      for (uint64_t ob = first + (uint64_t)lane; ob <= last; ob += 64) {
        // dest byte ob covers dest bits [ob*8, ob*8+8); src bit = dest - B
        int64_t sbit0 = (int64_t)(ob << 3) - (int64_t)B;  // may be -r
        uint32_t acc = 0;
        // gather 8 dest bits: src bits sbit0..sbit0+7 (clipped to [0,take))
        for (int j = 0; j < 8; j++) {
          int64_t sb = sbit0 + j;
          if (sb < 0 || sb >= (int64_t)take) continue;
          uint8_t sv = (p[pos + (uint32_t)(sb >> 3)] >> (sb & 7)) & 1;
          acc |= (uint32_t)sv << j;
        }
        if (acc) atomicOr((unsigned int*)(valid + (ob & ~3ull)),
                          (unsigned int)acc << ((ob & 3) * 8));
        local_ones += __popc(acc);
      }
      // wave-sum the ones counted by each lane
      for (int s = 32; s; s >>= 1)
        local_ones += __shfl_down(local_ones, s, 64);
      ones += __shfl(local_ones, 0, 64);
      (void)r;
      filled += take;
      pos += bytes;
    } else {  // repeated run
      uint32_t cnt = (uint32_t)(header >> 1);
      if (pos + 1 > len) {
        *bad = true;
        return 0;
      }
      uint8_t v = p[pos++];
      uint32_t take = cnt < count - filled ? cnt : count - filled;
      if (v) {
        uint64_t first = B >> 3, last = (B + take - 1) >> 3;
        for (uint64_t ob = first + (uint64_t)lane; ob <= last; ob += 64) {
          uint32_t lo = (uint64_t)(ob << 3) > B ? 0 : (uint32_t)(B & 7);
          uint64_t endbit = B + take;  // exclusive
          uint32_t hi = (ob << 3) + 8 <= endbit ? 8
                                                : (uint32_t)(endbit - (ob << 3));
          uint32_t mask = ((hi >= 8 ? 0xFFu : ((1u << hi) - 1)) &
                           ~((1u << lo) - 1));
          atomicOr((unsigned int*)(valid + (ob & ~3ull)),
                   (unsigned int)mask << ((ob & 3) * 8));
        }
        ones += take;
      }
      filled += take;
    }
  }
  return ones;
}

__global__ void __launch_bounds__(64) k_pq_pages_decode(
    const uint8_t* __restrict__ comp, const PqGpuPage* __restrict__ pages,
    int npages, uint8_t* __restrict__ scratch,
    uint8_t* __restrict__ valid_blob,
    const PqGpuChunk* __restrict__ chunks, uint32_t* __restrict__ nn_counts,
    uint32_t* __restrict__ val_offs, uint32_t* __restrict__ err) {
  int pg = (int)blockIdx.x;
  if (pg >= npages) return;
  const PqGpuPage p = pages[pg];
  const uint8_t* src = comp + p.comp_off;
  uint8_t* dst = scratch + p.uncomp_off;
  if (!dev_snappy_page(src, p.comp_len, dst, p.uncomp_len)) {
    if (threadIdx.x == 0) atomicOr(err, 1u);
    return;
  }
  uint32_t voff = 0;
  uint32_t nn = p.num_values;
  if (p.has_def) {
    if (p.uncomp_len < 4) {
      if (threadIdx.x == 0) atomicOr(err, 2u);
      return;
    }
    uint32_t ll = *(const uint32_t*)dst;
    if (ll > p.uncomp_len - 4) {
      if (threadIdx.x == 0) atomicOr(err, 2u);
      return;
    }
    bool bad = false;
    const PqGpuChunk c = chunks[p.chunk_id];
    nn = dev_def_walk(dst + 4, ll, p.num_values,
                      valid_blob + c.valid_base,
                      (uint64_t)p.value_base, &bad);
    if (bad) {
      if (threadIdx.x == 0) atomicOr(err, 2u);
      return;
    }
    voff = 4 + ll;
  }
  if (threadIdx.x == 0) {
    nn_counts[pg] = nn;
    val_offs[pg] = voff;
  }
}

// per-chunk exclusive scan of page nn counts -> dense page offsets (value
// units); also the chunk nn total. One block per chunk, pages are few.
__global__ void k_pq_page_offsets(const PqGpuChunk* __restrict__ chunks,
                                  int nchunks,
                                  const uint32_t* __restrict__ nn_counts,
                                  uint32_t* __restrict__ page_dense,
                                  uint32_t* __restrict__ chunk_nn) {
  int c = (int)blockIdx.x;
  if (c >= nchunks || threadIdx.x != 0) return;
  const PqGpuChunk ch = chunks[c];
  uint32_t acc = 0;
  for (uint32_t i = 0; i < ch.npages; i++) {
    page_dense[ch.page0 + i] = acc;
    acc += nn_counts[ch.page0 + i];
  }
  chunk_nn[c] = acc;
}

// compact each page's dense (non-null) value bytes into the chunk's dense
// blob region. 256 threads per page: pure streaming copy.
__global__ void __launch_bounds__(256) k_pq_pages_compact(
    const PqGpuPage* __restrict__ pages, int npages,
    const uint8_t* __restrict__ scratch, const PqGpuChunk* __restrict__ chunks,
    const uint32_t* __restrict__ nn_counts,
    const uint32_t* __restrict__ val_offs,
    const uint32_t* __restrict__ page_dense,
    uint8_t* __restrict__ dense_blob) {
  int pg = (int)blockIdx.x;
  if (pg >= npages) return;
  const PqGpuPage p = pages[pg];
  const PqGpuChunk c = chunks[p.chunk_id];
  const uint8_t* src = scratch + p.uncomp_off + val_offs[pg];
  size_t bytes = (size_t)nn_counts[pg] * c.vw;
  if (val_offs[pg] + bytes > p.uncomp_len) bytes = 0;  // err flag already set
  uint8_t* dst = dense_blob + c.dense_base +
                 ((size_t)c.prefix_nn + page_dense[pg]) * c.vw;
  // byte copy: src starts at 4+ll (arbitrary alignment); byte loads by 256
  // consecutive lanes coalesce into full lines
  for (size_t i = threadIdx.x; i < bytes; i += 256) dst[i] = src[i];
}

void launch_pq_pages_decode(const uint8_t* comp, const PqGpuPage* pages,
                            int npages, uint8_t* scratch, uint8_t* valid_blob,
                            const PqGpuChunk* chunks, uint32_t* nn_counts,
                            uint32_t* val_offs, uint32_t* err, hipStream_t s) {
  hipLaunchKernelGGL(k_pq_pages_decode, dim3(npages), dim3(64), 0, s, comp,
                     pages, npages, scratch, valid_blob, chunks, nn_counts,
                     val_offs, err);
  check_launch_pq("k_pq_pages_decode");
}

void launch_pq_page_offsets(const PqGpuChunk* chunks, int nchunks,
                            const uint32_t* nn_counts, uint32_t* page_dense,
                            uint32_t* chunk_nn, hipStream_t s) {
  hipLaunchKernelGGL(k_pq_page_offsets, dim3(nchunks), dim3(64), 0, s, chunks,
                     nchunks, nn_counts, page_dense, chunk_nn);
  check_launch_pq("k_pq_page_offsets");
}

void launch_pq_pages_compact(const PqGpuPage* pages, int npages,
                             const uint8_t* scratch, const PqGpuChunk* chunks,
                             const uint32_t* nn_counts,
                             const uint32_t* val_offs,
                             const uint32_t* page_dense,
                             uint8_t* dense_blob, hipStream_t s) {
  hipLaunchKernelGGL(k_pq_pages_compact, dim3(npages), dim3(256), 0, s, pages,
                     npages, scratch, chunks, nn_counts, val_offs, page_dense,
                     dense_blob);
  check_launch_pq("k_pq_pages_compact");
}

}  // namespace auron
