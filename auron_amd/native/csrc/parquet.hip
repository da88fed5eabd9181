// GPU Parquet page decode — gfx950.
//
// Role parity: the reference's CPU parquet decode (parquet_exec.rs via
// arrow-rs). Here the host ships raw uncompressed column-chunk bytes to
// HBM and these kernels do the decode work:
//  - k_pq_rle1:      RLE/bit-packed-hybrid definition levels (bit width 1)
//                    -> validity bytes, one workgroup per page
//  - k_pq_scatter:   PLAIN non-null values scattered to row slots using a
//                    precomputed validity prefix sum
//  - k_pq_copy:      PLAIN values for required (no-null) chunks, segmented
//                    vectorized copy
//
// Page descriptor layout (int64 x 6, packed by parquet_native.py):
//   [0]=def_off [1]=def_len [2]=values_off [3]=n_values [4]=row_start [5]=pad

#include <hip/hip_runtime.h>
#include <stdint.h>

#define AU_EXPORT extern "C" __attribute__((visibility("default")))

struct PqPage {
  int64_t def_off;
  int64_t def_len;
  int64_t values_off;
  int64_t n_values;
  int64_t row_start;
  int64_t pad;
};

// ------------------------------------------------------------------- RLE1
// Two-phase per block: lane 0 parses run headers into an LDS window, then
// all 256 threads expand the window in parallel. Runs are (value,count) or
// bit-packed literal groups.
struct Run {
  int32_t out_start;  // relative to page
  int32_t count;
  int32_t src_off;    // for literal runs: byte offset of packed bits
  int32_t rep_val;    // for repeat runs: value; -1 => literal
};

#define RLE_WINDOW 512

// Split factor: each page gets RLE1_SPLIT workgroups. Writers (pyarrow,
// parquet-mr) emit definition levels as ONE hybrid run per page in the
// common case; that run is sliced across the page's workgroups so the
// whole chip fills even for a handful of pages. Multi-run pages fall back
// to the windowed serial-parse path in workgroup 0.
#define RLE1_SPLIT 16

#define RLE1_LDS_BYTES 16384

__global__ void k_pq_rle1(const PqPage* pages, const uint8_t* buf, uint8_t* out) {
  const PqPage p = pages[blockIdx.x / RLE1_SPLIT];
  const int slice = blockIdx.x % RLE1_SPLIT;
  const uint8_t* src = buf + p.def_off;
  int64_t src_len = p.def_len;
  uint8_t* dst = out + p.row_start;
  const int64_t n = p.n_values;

  // Stage the page's level bytes through LDS: the run-header parse is a
  // serial chain of dependent byte loads, and HBM-latency per byte is
  // what made this kernel the r2 profile's #1 entry. A bit-width-1 page
  // of 8k values is ~1-3 KB, far under the 160 KB LDS per CU.
  __shared__ uint8_t sbuf[RLE1_LDS_BYTES];
  if (src_len <= RLE1_LDS_BYTES) {
    for (int64_t i = threadIdx.x; i < src_len; i += blockDim.x)
      sbuf[i] = src[i];
    __syncthreads();
    src = sbuf;
  }

  // ---- fast path: single run spans the page ----
  {
    int64_t pos = 0;
    uint64_t header = 0;
    int shift = 0;
    while (pos < src_len) {
      uint8_t b = src[pos++];
      header |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    int64_t covered = (header & 1) ? (int64_t)(header >> 1) * 8
                                   : (int64_t)(header >> 1);
    if (covered >= n) {
      const int64_t per = (n + RLE1_SPLIT - 1) / RLE1_SPLIT;
      const int64_t lo = slice * per;
      const int64_t hi = min(lo + per, n);
      if (lo >= hi) return;
      if (header & 1) {
        const uint8_t* bits = src + pos;
        for (int64_t j = lo + threadIdx.x; j < hi; j += blockDim.x)
          dst[j] = (bits[j >> 3] >> (j & 7)) & 1;
      } else {
        const uint8_t v = src[pos] & 1;
        for (int64_t j = lo + threadIdx.x; j < hi; j += blockDim.x)
          dst[j] = v;
      }
      return;
    }
  }
  if (slice != 0) return;  // general path: one workgroup per page

  __shared__ Run runs[RLE_WINDOW];
  __shared__ int nruns;
  __shared__ int64_t s_pos, s_i;

  if (threadIdx.x == 0) {
    s_pos = 0;
    s_i = 0;
  }
  __syncthreads();

  while (true) {
    if (threadIdx.x == 0) {
      int64_t pos = s_pos;
      int64_t i = s_i;
      int r = 0;
      while (r < RLE_WINDOW && i < n && pos < src_len) {
        // uvarint
        uint64_t header = 0;
        int shift = 0;
        while (true) {
          uint8_t b = src[pos++];
          header |= (uint64_t)(b & 0x7F) << shift;
          if (!(b & 0x80)) break;
          shift += 7;
        }
        if (header & 1) {  // literal: (header>>1) groups of 8 bit-packed
          int64_t ngroups = header >> 1;
          int64_t nvals = ngroups * 8;
          if (nvals > n - i) nvals = n - i;
          runs[r].out_start = (int32_t)i;
          runs[r].count = (int32_t)nvals;
          runs[r].src_off = (int32_t)pos;
          runs[r].rep_val = -1;
          pos += ngroups;
          i += nvals;
        } else {
          int64_t cnt = header >> 1;
          if (cnt > n - i) cnt = n - i;
          runs[r].out_start = (int32_t)i;
          runs[r].count = (int32_t)cnt;
          runs[r].rep_val = src[pos] & 1;
          pos += 1;
          i += cnt;
        }
        r++;
      }
      nruns = r;
      s_pos = pos;
      s_i = i;
    }
    __syncthreads();
    int count = nruns;
    if (count == 0) break;
    // expand: threads stride over runs
    for (int r = threadIdx.x / 64; r < count; r += blockDim.x / 64) {
      const Run run = runs[r];
      int lane = threadIdx.x & 63;
      if (run.rep_val >= 0) {
        for (int j = lane; j < run.count; j += 64) dst[run.out_start + j] = (uint8_t)run.rep_val;
      } else {
        for (int j = lane; j < run.count; j += 64) {
          dst[run.out_start + j] = (src[run.src_off + (j >> 3)] >> (j & 7)) & 1;
        }
      }
    }
    __syncthreads();
    if (count < RLE_WINDOW) break;
  }
}

// --------------------------------------------------- host-parsed RLE1 path
// The run-header walk is inherently serial PER PAGE and, at ~2-3k runs per
// page for a few-percent-null column, dominates decode time if done on one
// device thread. Pages are static per file, so the HOST parses headers once
// (this plain-C routine, cached in the reader's file metadata) and the
// device only runs the embarrassingly-parallel expansion kernel below.
// Run encoding: out_start/count are chunk-absolute rows; src_off is an
// absolute byte offset into the staged buffer; rep_val -1 = literal run.
AU_EXPORT int64_t au_host_rle1_parse(const uint8_t* src, int64_t src_len,
                                     int64_t n, int64_t out_base,
                                     int64_t src_base, int32_t* runs,
                                     int64_t cap) {
  int64_t pos = 0, i = 0, r = 0;
  while (i < n && pos < src_len) {
    uint64_t header = 0;
    int shift = 0;
    while (pos < src_len) {
      uint8_t b = src[pos++];
      header |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    if (r >= cap) return -1;
    if (header & 1) {
      int64_t ngroups = (int64_t)(header >> 1);
      int64_t nvals = ngroups * 8;
      if (nvals > n - i) nvals = n - i;
      runs[4 * r + 0] = (int32_t)(out_base + i);
      runs[4 * r + 1] = (int32_t)nvals;
      runs[4 * r + 2] = (int32_t)(src_base + pos);
      runs[4 * r + 3] = -1;
      pos += ngroups;
      i += nvals;
    } else {
      int64_t cnt = (int64_t)(header >> 1);
      if (cnt > n - i) cnt = n - i;
      runs[4 * r + 0] = (int32_t)(out_base + i);
      runs[4 * r + 1] = (int32_t)cnt;
      runs[4 * r + 2] = 0;
      runs[4 * r + 3] = src[pos] & 1;
      pos += 1;
      i += cnt;
    }
    r++;
  }
  return i == n ? r : -1;
}

__global__ void k_rle1_expand(const int32_t* __restrict__ runs, int64_t nruns,
                              const uint8_t* __restrict__ buf,
                              uint8_t* __restrict__ out) {
  const int64_t wave = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t nwaves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  for (int64_t r = wave; r < nruns; r += nwaves) {
    const int32_t out_start = runs[4 * r + 0];
    const int32_t count = runs[4 * r + 1];
    const int32_t src_off = runs[4 * r + 2];
    const int32_t rep = runs[4 * r + 3];
    if (rep >= 0) {
      const uint8_t v = (uint8_t)rep;
      for (int32_t j = lane; j < count; j += 64) out[out_start + j] = v;
    } else {
      const uint8_t* bits = buf + src_off;
      for (int32_t j = lane; j < count; j += 64)
        out[out_start + j] = (bits[j >> 3] >> (j & 7)) & 1;
    }
  }
}

AU_EXPORT int au_rle1_expand(const int32_t* runs_dev, int64_t nruns,
                             const void* buf, uint8_t* out, void* stream) {
  if (nruns == 0) return 0;
  int64_t blocks = (nruns * 64 + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  hipLaunchKernelGGL(k_rle1_expand, dim3((uint32_t)blocks), dim3(256), 0,
                     (hipStream_t)stream, runs_dev, nruns,
                     (const uint8_t*)buf, out);
  return (int)hipGetLastError();
}

AU_EXPORT int au_pq_rle1(const void* pages_dev, int npages, const void* buf,
                         uint8_t* out, void* stream) {
  if (npages == 0) return 0;
  hipLaunchKernelGGL(k_pq_rle1, dim3(npages * RLE1_SPLIT), dim3(256), 0,
                     (hipStream_t)stream,
                     (const PqPage*)pages_dev, (const uint8_t*)buf, out);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------------ RLE idx
// Generalized RLE/bit-packed hybrid decode for DICTIONARY indices
// (bit width 1..32). One workgroup per page; same two-phase structure as
// k_pq_rle1. Descriptor reuses PqPage with repurposed fields:
//   def_off=idx_off  def_len=idx_len  values_off=unused
//   n_values=page rows  row_start=chunk-relative row  pad=bit_width
// `prefix` is the chunk-wide inclusive cumsum of validity (or null for
// no-null chunks): the kernel derives each page's valid count and its
// compact output base on-device, so the host never syncs.
// The staged buffer is padded by 8 bytes so the unaligned 8-byte loads at
// a literal run's tail never read out of bounds.
__global__ void k_pq_rle_idx(const PqPage* pages, const uint8_t* buf, int32_t* out,
                             const int64_t* prefix) {
  const PqPage p = pages[blockIdx.x];
  const uint8_t* src = buf + p.def_off;
  const int64_t src_len = p.def_len;
  int64_t base, n;
  if (prefix) {
    base = p.row_start == 0 ? 0 : prefix[p.row_start - 1];
    n = prefix[p.row_start + p.n_values - 1] - base;
  } else {
    base = p.row_start;
    n = p.n_values;
  }
  int32_t* dst = out + base;
  const int bw = (int)p.pad;
  const uint64_t mask = bw >= 64 ? ~0ull : ((1ull << bw) - 1);
  const int vbytes = (bw + 7) >> 3;

  __shared__ Run runs[RLE_WINDOW];
  __shared__ int nruns;
  __shared__ int64_t s_pos, s_i;

  if (threadIdx.x == 0) {
    s_pos = 0;
    s_i = 0;
  }
  __syncthreads();

  while (true) {
    if (threadIdx.x == 0) {
      int64_t pos = s_pos;
      int64_t i = s_i;
      int r = 0;
      while (r < RLE_WINDOW && i < n && pos < src_len) {
        uint64_t header = 0;
        int shift = 0;
        while (true) {
          uint8_t b = src[pos++];
          header |= (uint64_t)(b & 0x7F) << shift;
          if (!(b & 0x80)) break;
          shift += 7;
        }
        if (header & 1) {  // literal: (header>>1) groups of 8, bw bits each
          int64_t ngroups = header >> 1;
          int64_t nvals = ngroups * 8;
          if (nvals > n - i) nvals = n - i;
          runs[r].out_start = (int32_t)i;
          runs[r].count = (int32_t)nvals;
          runs[r].src_off = (int32_t)pos;
          runs[r].rep_val = -1;
          pos += ngroups * bw;  // bw bytes per group of 8
          i += nvals;
        } else {
          int64_t cnt = header >> 1;
          if (cnt > n - i) cnt = n - i;
          uint32_t v = 0;
          for (int b2 = 0; b2 < vbytes; b2++) v |= (uint32_t)src[pos + b2] << (8 * b2);
          runs[r].out_start = (int32_t)i;
          runs[r].count = (int32_t)cnt;
          runs[r].rep_val = (int32_t)(v & mask);
          pos += vbytes;
          i += cnt;
        }
        r++;
      }
      nruns = r;
      s_pos = pos;
      s_i = i;
    }
    __syncthreads();
    int count = nruns;
    if (count == 0) break;
    for (int r = threadIdx.x / 64; r < count; r += blockDim.x / 64) {
      const Run run = runs[r];
      int lane = threadIdx.x & 63;
      if (run.rep_val >= 0) {
        for (int j = lane; j < run.count; j += 64) dst[run.out_start + j] = run.rep_val;
      } else {
        for (int j = lane; j < run.count; j += 64) {
          int64_t bitpos = (int64_t)j * bw;
          uint64_t w;
          __builtin_memcpy(&w, src + run.src_off + (bitpos >> 3), 8);
          dst[run.out_start + j] = (int32_t)((w >> (bitpos & 7)) & mask);
        }
      }
    }
    __syncthreads();
    if (count < RLE_WINDOW) break;
  }
}

AU_EXPORT int au_pq_rle_idx(const void* pages_dev, int npages, const void* buf,
                            int32_t* out, const int64_t* prefix, void* stream) {
  if (npages == 0) return 0;
  hipLaunchKernelGGL(k_pq_rle_idx, dim3(npages), dim3(256), 0, (hipStream_t)stream,
                     (const PqPage*)pages_dev, (const uint8_t*)buf, out, prefix);
  return (int)hipGetLastError();
}

// ----------------------------------------------------------------- scatter
// out[row] = valid(row) ? values[prefix[row]-1 - page_base] : 0
// prefix = inclusive cumsum of validity over the whole chunk (int64).
template <typename T>
__global__ void k_pq_scatter(const PqPage* pages, int npages, const uint8_t* buf,
                             const uint8_t* validity, const int64_t* prefix,
                             T* out, int64_t total) {
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < total;
       row += (int64_t)gridDim.x * blockDim.x) {
    // binary search the page containing `row`
    int lo = 0, hi = npages - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (pages[mid].row_start <= row) lo = mid; else hi = mid - 1;
    }
    const PqPage p = pages[lo];
    if (!validity[row]) {
      out[row] = (T)0;
      continue;
    }
    int64_t page_base = p.row_start == 0 ? 0 : prefix[p.row_start - 1];
    int64_t k = prefix[row] - 1 - page_base;
    T v;
    __builtin_memcpy(&v, buf + p.values_off + k * (int64_t)sizeof(T), sizeof(T));
    out[row] = v;
  }
}

// ---------------------------------------------------------------- copy (no nulls)
template <typename T>
__global__ void k_pq_copy(const PqPage* pages, int npages, const uint8_t* buf,
                          T* out, int64_t total) {
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < total;
       row += (int64_t)gridDim.x * blockDim.x) {
    int lo = 0, hi = npages - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (pages[mid].row_start <= row) lo = mid; else hi = mid - 1;
    }
    const PqPage p = pages[lo];
    int64_t k = row - p.row_start;
    T v;
    __builtin_memcpy(&v, buf + p.values_off + k * (int64_t)sizeof(T), sizeof(T));
    out[row] = v;
  }
}

static inline int pq_grid(int64_t n) {
  int64_t g = (n + 255) / 256;
  if (g > 2048) g = 2048;
  if (g < 1) g = 1;
  return (int)g;
}

AU_EXPORT int au_pq_scatter(const void* pages_dev, int npages, const void* buf,
                            const uint8_t* validity, const int64_t* prefix,
                            void* out, int esize, int64_t total, void* stream) {
  if (total == 0) return 0;
  hipStream_t s = (hipStream_t)stream;
  dim3 g(pq_grid(total)), b(256);
  if (esize == 4)
    hipLaunchKernelGGL((k_pq_scatter<uint32_t>), g, b, 0, s, (const PqPage*)pages_dev,
                       npages, (const uint8_t*)buf, validity, prefix, (uint32_t*)out, total);
  else
    hipLaunchKernelGGL((k_pq_scatter<uint64_t>), g, b, 0, s, (const PqPage*)pages_dev,
                       npages, (const uint8_t*)buf, validity, prefix, (uint64_t*)out, total);
  return (int)hipGetLastError();
}

AU_EXPORT int au_pq_copy_plain(const void* pages_dev, int npages, const void* buf,
                               void* out, int esize, int64_t total, void* stream) {
  if (total == 0) return 0;
  hipStream_t s = (hipStream_t)stream;
  dim3 g(pq_grid(total)), b(256);
  if (esize == 4)
    hipLaunchKernelGGL((k_pq_copy<uint32_t>), g, b, 0, s, (const PqPage*)pages_dev,
                       npages, (const uint8_t*)buf, (uint32_t*)out, total);
  else
    hipLaunchKernelGGL((k_pq_copy<uint64_t>), g, b, 0, s, (const PqPage*)pages_dev,
                       npages, (const uint8_t*)buf, (uint64_t*)out, total);
  return (int)hipGetLastError();
}

// Host-side walk of a PLAIN BYTE_ARRAY data-page payload:
// [u32 len][bytes][u32 len][bytes]... -> per-value (absolute offset, length).
// Returns the number of values parsed, or -1 on overflow/corruption.
// (arrow-rs does the equivalent offset materialization host-side too;
// the variable-length walk is inherently serial.)
AU_EXPORT int64_t au_host_plainba_parse(const uint8_t* buf, int64_t off,
                                        int64_t len, int64_t* offs,
                                        int32_t* lens, int64_t cap) {
  int64_t pos = off, end = off + len, n = 0;
  while (pos + 4 <= end) {
    uint32_t l = (uint32_t)buf[pos] | ((uint32_t)buf[pos + 1] << 8) |
                 ((uint32_t)buf[pos + 2] << 16) | ((uint32_t)buf[pos + 3] << 24);
    pos += 4;
    if ((int64_t)l > end - pos || n >= cap) return -1;
    offs[n] = pos;
    lens[n] = (int32_t)l;
    pos += l;
    n++;
  }
  return n;
}

// Host-side DELTA_BINARY_PACKED decode (parquet encoding 5) into PLAIN
// little-endian values (esize 4 or 8). The format is inherently serial
// (varint headers, running prefix value); decoded pages then ride the
// same device PLAIN path as everything else. Returns values written,
// -1 on corruption.
static inline int64_t du_uvarint(const uint8_t* b, int64_t end, int64_t* pos,
                                 uint64_t* out) {
  uint64_t v = 0; int shift = 0;
  while (*pos < end) {
    uint8_t x = b[(*pos)++];
    v |= (uint64_t)(x & 0x7F) << shift;
    if (!(x & 0x80)) { *out = v; return 0; }
    shift += 7;
  }
  return -1;
}

AU_EXPORT int64_t au_host_delta_unpack(const uint8_t* buf, int64_t off,
                                       int64_t len, int64_t n, int esize,
                                       uint8_t* out) {
  int64_t pos = off, end = off + len;
  uint64_t block_size, nmini, total, ufirst;
  if (du_uvarint(buf, end, &pos, &block_size)) return -1;
  if (du_uvarint(buf, end, &pos, &nmini)) return -1;
  if (du_uvarint(buf, end, &pos, &total)) return -1;
  if (du_uvarint(buf, end, &pos, &ufirst)) return -1;
  if (nmini == 0 || block_size % nmini) return -1;
  // `n` is the output CAPACITY (page row count); the page encodes
  // `total` values (= present values when the column has nulls)
  if ((int64_t)total < n) n = (int64_t)total;
  if (n <= 0) return 0;
  int64_t per_mini = (int64_t)(block_size / nmini);
  int64_t value = (int64_t)((ufirst >> 1) ^ -(int64_t)(ufirst & 1));
  int64_t count = 0;
  #define EMIT(v) do { \
    if (esize == 4) { int32_t t = (int32_t)(v); __builtin_memcpy(out + 4 * count, &t, 4); } \
    else { int64_t t = (v); __builtin_memcpy(out + 8 * count, &t, 8); } \
    count++; } while (0)
  EMIT(value);
  while (count < n) {
    uint64_t umin;
    if (du_uvarint(buf, end, &pos, &umin)) return -1;
    int64_t min_delta = (int64_t)((umin >> 1) ^ -(int64_t)(umin & 1));
    if (pos + (int64_t)nmini > end) return -1;
    const uint8_t* bws = buf + pos;
    pos += nmini;
    for (uint64_t mb = 0; mb < nmini && count < n; mb++) {
      int bw = bws[mb];
      if (bw > 64) return -1;
      int64_t nbytes = (per_mini * bw + 7) / 8;
      if (pos + nbytes > end) return -1;
      // LSB-first bit unpack
      int64_t bit = 0;
      for (int64_t i = 0; i < per_mini && count < n; i++) {
        uint64_t d = 0;
        for (int got = 0; got < bw; ) {
          int64_t byte_i = pos + (bit >> 3);
          int within = (int)(bit & 7);
          int take = 8 - within;
          if (take > bw - got) take = bw - got;
          uint64_t bits = ((uint64_t)buf[byte_i] >> within) & ((1ULL << take) - 1);
          d |= bits << got;
          got += take;
          bit += take;
        }
        value += min_delta + (int64_t)d;
        EMIT(value);
      }
      pos += nbytes;
    }
  }
  #undef EMIT
  return count;
}
