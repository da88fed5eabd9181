// Auron-AMD native engine kernels — gfx950 (MI355X, CDNA4) only.
//
// Role parity (reference, for capability mapping — no code is ported):
//  - Spark-compatible murmur3 row hash:   /root/reference/native-engine/datafusion-ext-commons/src/spark_hash.rs
//  - group-by hash table:                 .../datafusion-ext-plans/src/agg/agg_hash_map.rs
//  - join hash table build/probe:         .../datafusion-ext-plans/src/joins/join_hash_map.rs
//
// Design: MI355X-first. 64-wide wavefronts, grid-stride loops capped at
// ~2048 workgroups (256 CUs x 8 blocks — fills all 8 XCDs), device-scope
// atomics for cross-XCD correctness (per-XCD L2s are not coherent).
// Tables live in HBM3E; LDS-staged variants are layered on top where
// profiling shows payoff.
//
// C ABI only — loaded via ctypes from auron_amd/native/__init__.py.
// All pointers are device pointers owned by torch; `stream` is the torch
// current stream. No allocation happens here.

#include <hip/hip_runtime.h>
#include <stdint.h>

#define AU_EXPORT extern "C" __attribute__((visibility("default")))

// dtype codes — keep in sync with auron_amd/dtypes.py
enum AuDType : int32_t {
  AU_BOOL = 0,
  AU_INT8 = 1,
  AU_INT16 = 2,
  AU_INT32 = 3,
  AU_INT64 = 4,
  AU_FLOAT32 = 5,
  AU_FLOAT64 = 6,
  AU_DATE32 = 7,
  AU_STRING = 8,
  AU_DECIMAL64 = 9,
  AU_LIST = 10,
  AU_TIMESTAMP = 11,  // int64 microseconds since epoch
};

struct AuColDesc {
  const void* data;        // typed buffer; for STRING: uint8 bytes
  const int64_t* offsets;  // STRING only, n+1 (LargeString: 64-bit)
  const uint8_t* validity; // bool bytes (1 = valid) or nullptr
  int32_t dtype;
  int32_t scale;
};

static inline int au_grid(int64_t n, int block) {
  int64_t g = (n + block - 1) / block;
  if (g > 2048) g = 2048;  // 256 CU x 8 — grid-stride the rest
  if (g < 1) g = 1;
  return (int)g;
}

// ----------------------------------------------------------------- murmur3
// Spark Murmur3_x86_32 (seed-chained across columns, null-skipping).
__device__ __forceinline__ uint32_t rotl32(uint32_t x, int r) {
  return (x << r) | (x >> (32 - r));
}
__device__ __forceinline__ uint32_t mixK1(uint32_t k1) {
  k1 *= 0xcc9e2d51u;
  k1 = rotl32(k1, 15);
  k1 *= 0x1b873593u;
  return k1;
}
__device__ __forceinline__ uint32_t mixH1(uint32_t h1, uint32_t k1) {
  h1 ^= k1;
  h1 = rotl32(h1, 13);
  h1 = h1 * 5u + 0xe6546b64u;
  return h1;
}
__device__ __forceinline__ uint32_t fmix(uint32_t h1, uint32_t len) {
  h1 ^= len;
  h1 ^= h1 >> 16;
  h1 *= 0x85ebca6bu;
  h1 ^= h1 >> 13;
  h1 *= 0xc2b2ae35u;
  h1 ^= h1 >> 16;
  return h1;
}
__device__ __forceinline__ uint32_t hash_int(uint32_t v, uint32_t seed) {
  return fmix(mixH1(seed, mixK1(v)), 4);
}
__device__ __forceinline__ uint32_t hash_long(uint64_t v, uint32_t seed) {
  uint32_t h1 = mixH1(seed, mixK1((uint32_t)v));
  h1 = mixH1(h1, mixK1((uint32_t)(v >> 32)));
  return fmix(h1, 8);
}
__device__ uint32_t hash_bytes(const uint8_t* p, int32_t len, uint32_t seed) {
  int32_t aligned = len & ~3;
  uint32_t h1 = seed;
  for (int32_t i = 0; i < aligned; i += 4) {
    uint32_t w = (uint32_t)p[i] | ((uint32_t)p[i + 1] << 8) |
                 ((uint32_t)p[i + 2] << 16) | ((uint32_t)p[i + 3] << 24);
    h1 = mixH1(h1, mixK1(w));
  }
  for (int32_t i = aligned; i < len; i++) {
    // Spark sign-extends each tail byte to int
    int32_t b = (int8_t)p[i];
    h1 = mixH1(h1, mixK1((uint32_t)b));
  }
  return fmix(h1, (uint32_t)len);
}

__device__ __forceinline__ bool row_valid(const AuColDesc& c, int64_t i) {
  return c.validity == nullptr || c.validity[i] != 0;
}

__device__ uint32_t hash_one(const AuColDesc& c, int64_t i, uint32_t seed) {
  switch (c.dtype) {
    case AU_BOOL:
      return hash_int(((const uint8_t*)c.data)[i] ? 1u : 0u, seed);
    case AU_INT8:
      return hash_int((uint32_t)(int32_t)((const int8_t*)c.data)[i], seed);
    case AU_INT16:
      return hash_int((uint32_t)(int32_t)((const int16_t*)c.data)[i], seed);
    case AU_INT32:
    case AU_DATE32:
      return hash_int((uint32_t)((const int32_t*)c.data)[i], seed);
    case AU_INT64:
    case AU_DECIMAL64:
    case AU_TIMESTAMP:
      return hash_long((uint64_t)((const int64_t*)c.data)[i], seed);
    case AU_FLOAT32: {
      float f = ((const float*)c.data)[i];
      if (f == 0.0f) f = 0.0f;  // normalize -0.0
      return hash_int(__float_as_uint(f), seed);
    }
    case AU_FLOAT64: {
      double d = ((const double*)c.data)[i];
      if (d == 0.0) d = 0.0;
      return hash_long((uint64_t)__double_as_longlong(d), seed);
    }
    case AU_STRING: {
      int64_t s = c.offsets[i];
      int64_t e = c.offsets[i + 1];
      return hash_bytes((const uint8_t*)c.data + s, (int32_t)(e - s), seed);
    }
  }
  return seed;
}

__global__ void k_murmur3(const AuColDesc* cols, int ncols, int64_t n,
                          uint32_t seed, int32_t* out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t h = seed;
    for (int c = 0; c < ncols; c++) {
      if (row_valid(cols[c], i)) h = hash_one(cols[c], i, h);
    }
    out[i] = (int32_t)h;
  }
}

AU_EXPORT int au_murmur3(const void* cols_dev, int ncols, int64_t n,
                         int32_t seed, int32_t* out, void* stream) {
  if (n == 0) return 0;
  hipLaunchKernelGGL(k_murmur3, dim3(au_grid(n, 256)), dim3(256), 0,
                     (hipStream_t)stream, (const AuColDesc*)cols_dev, ncols, n,
                     (uint32_t)seed, out);
  return (int)hipGetLastError();
}

// -------------------------------------------------------------- key equality
// Rows may come from two different batches (probe vs build).
__device__ bool keys_equal(const AuColDesc* a_cols, int64_t ai,
                           const AuColDesc* b_cols, int64_t bi, int ncols) {
  for (int c = 0; c < ncols; c++) {
    const AuColDesc& a = a_cols[c];
    const AuColDesc& b = b_cols[c];
    bool av = row_valid(a, ai);
    bool bv = row_valid(b, bi);
    if (av != bv) return false;
    if (!av) continue;  // both null: equal for GROUP BY grouping
    switch (a.dtype) {
      case AU_BOOL:
      case AU_INT8:
        if (((const int8_t*)a.data)[ai] != ((const int8_t*)b.data)[bi]) return false;
        break;
      case AU_INT16:
        if (((const int16_t*)a.data)[ai] != ((const int16_t*)b.data)[bi]) return false;
        break;
      case AU_INT32:
      case AU_DATE32:
        if (((const int32_t*)a.data)[ai] != ((const int32_t*)b.data)[bi]) return false;
        break;
      case AU_INT64:
      case AU_DECIMAL64:
      case AU_TIMESTAMP:
        if (((const int64_t*)a.data)[ai] != ((const int64_t*)b.data)[bi]) return false;
        break;
      case AU_FLOAT32:
        if (((const float*)a.data)[ai] != ((const float*)b.data)[bi]) return false;
        break;
      case AU_FLOAT64:
        if (((const double*)a.data)[ai] != ((const double*)b.data)[bi]) return false;
        break;
      case AU_STRING: {
        int64_t as = a.offsets[ai], ae = a.offsets[ai + 1];
        int64_t bs = b.offsets[bi], be = b.offsets[bi + 1];
        if (ae - as != be - bs) return false;
        const uint8_t* ap = (const uint8_t*)a.data + as;
        const uint8_t* bp = (const uint8_t*)b.data + bs;
        int32_t len = (int32_t)(ae - as);
        int32_t k = 0;
        for (; k + 8 <= len; k += 8) {
          uint64_t wa, wb;
          __builtin_memcpy(&wa, ap + k, 8);
          __builtin_memcpy(&wb, bp + k, 8);
          if (wa != wb) return false;
        }
        for (; k < len; k++)
          if (ap[k] != bp[k]) return false;
        break;
      }
    }
  }
  return true;
}

// ------------------------------------------------------------ group-by table
// Open-addressing, linear probing. slots[cap] holds representative row idx
// (-1 empty). cap is a power of two >= 2*n_distinct-safe (caller sizes 2n).
__global__ void k_group_insert(const AuColDesc* cols, int ncols, int64_t n,
                               int32_t* slots, uint32_t cap_mask,
                               const int32_t* hashes, int32_t* rep) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t p = ((uint32_t)hashes[i]) & cap_mask;
    int32_t me = (int32_t)i;
    while (true) {
      int32_t cur = slots[p];
      if (cur == -1) {
        int32_t prev = atomicCAS(&slots[p], -1, me);
        if (prev == -1) {
          rep[i] = me;
          break;
        }
        cur = prev;
      }
      if (keys_equal(cols, cur, cols, i, ncols)) {
        rep[i] = cur;
        break;
      }
      p = (p + 1) & cap_mask;
    }
  }
}

__global__ void k_group_assign(const int32_t* rep, int64_t n, int32_t* gid_of_row,
                               int64_t* rep_rows, int32_t* counter) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (rep[i] == (int32_t)i) {
      int32_t g = atomicAdd(counter, 1);
      gid_of_row[i] = g;
      rep_rows[g] = i;
    }
  }
}

__global__ void k_group_gather(const int32_t* rep, int64_t n, int32_t* gid_of_row) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    gid_of_row[i] = gid_of_row[rep[i]];
  }
}

AU_EXPORT int au_group_ids(const void* cols_dev, int ncols, int64_t n,
                           int32_t* slots, int64_t cap, const int32_t* hashes,
                           int32_t* rep, int32_t* gids, int64_t* rep_rows,
                           int32_t* counter, void* stream) {
  if (n == 0) return 0;
  hipStream_t s = (hipStream_t)stream;
  uint32_t mask = (uint32_t)(cap - 1);
  int g = au_grid(n, 256);
  hipLaunchKernelGGL(k_group_insert, dim3(g), dim3(256), 0, s,
                     (const AuColDesc*)cols_dev, ncols, n, slots, mask, hashes, rep);
  hipLaunchKernelGGL(k_group_assign, dim3(g), dim3(256), 0, s, rep, n, gids,
                     rep_rows, counter);
  hipLaunchKernelGGL(k_group_gather, dim3(g), dim3(256), 0, s, rep, n, gids);
  return (int)hipGetLastError();
}

// --------------------------------------------------------------- join table
// Chained buckets: heads[cap] (-1 empty) + next[n_build].
__global__ void k_join_build(int64_t n, int32_t* heads, uint32_t cap_mask,
                             int32_t* next, const int32_t* hashes) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t p = ((uint32_t)hashes[i]) & cap_mask;
    next[i] = atomicExch(&heads[p], (int32_t)i);
  }
}

// has_null_key: rows whose key contains a SQL NULL never match (unlike
// GROUP BY where nulls group together).
__device__ __forceinline__ bool join_row_has_null(const AuColDesc* cols,
                                                  int ncols, int64_t i) {
  for (int c = 0; c < ncols; c++)
    if (!row_valid(cols[c], i)) return true;
  return false;
}

__global__ void k_join_count(const AuColDesc* bcols, const AuColDesc* pcols,
                             int ncols, int64_t n_probe, const int32_t* heads,
                             uint32_t cap_mask, const int32_t* next,
                             const int32_t* phashes, int32_t* counts) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_probe;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t cnt = 0;
    if (!join_row_has_null(pcols, ncols, i)) {
      int32_t j = heads[((uint32_t)phashes[i]) & cap_mask];
      while (j != -1) {
        if (keys_equal(bcols, j, pcols, i, ncols) &&
            !join_row_has_null(bcols, ncols, j))
          cnt++;
        j = next[j];
      }
    }
    counts[i] = cnt;
  }
}

__global__ void k_join_fill(const AuColDesc* bcols, const AuColDesc* pcols,
                            int ncols, int64_t n_probe, const int32_t* heads,
                            uint32_t cap_mask, const int32_t* next,
                            const int32_t* phashes, const int64_t* offsets,
                            int64_t* build_out, int64_t* probe_out,
                            uint8_t* build_matched, int emit_unmatched_probe) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_probe;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t o = offsets[i];
    int64_t start = o;
    if (!join_row_has_null(pcols, ncols, i)) {
      int32_t j = heads[((uint32_t)phashes[i]) & cap_mask];
      while (j != -1) {
        if (keys_equal(bcols, j, pcols, i, ncols) &&
            !join_row_has_null(bcols, ncols, j)) {
          build_out[o] = j;
          probe_out[o] = i;
          if (build_matched) build_matched[j] = 1;
          o++;
        }
        j = next[j];
      }
    }
    if (emit_unmatched_probe && o == start) {
      build_out[o] = -1;
      probe_out[o] = i;
    }
  }
}

AU_EXPORT int au_join_build(int64_t n, int32_t* heads, int64_t cap,
                            int32_t* next, const int32_t* hashes, void* stream) {
  if (n == 0) return 0;
  hipLaunchKernelGGL(k_join_build, dim3(au_grid(n, 256)), dim3(256), 0,
                     (hipStream_t)stream, n, heads, (uint32_t)(cap - 1), next,
                     hashes);
  return (int)hipGetLastError();
}

AU_EXPORT int au_join_count(const void* bcols, const void* pcols, int ncols,
                            int64_t n_probe, const int32_t* heads, int64_t cap,
                            const int32_t* next, const int32_t* phashes,
                            int32_t* counts, void* stream) {
  if (n_probe == 0) return 0;
  hipLaunchKernelGGL(k_join_count, dim3(au_grid(n_probe, 256)), dim3(256), 0,
                     (hipStream_t)stream, (const AuColDesc*)bcols,
                     (const AuColDesc*)pcols, ncols, n_probe, heads,
                     (uint32_t)(cap - 1), next, phashes, counts);
  return (int)hipGetLastError();
}

AU_EXPORT int au_join_fill(const void* bcols, const void* pcols, int ncols,
                           int64_t n_probe, const int32_t* heads, int64_t cap,
                           const int32_t* next, const int32_t* phashes,
                           const int64_t* offsets, int64_t* build_out,
                           int64_t* probe_out, uint8_t* build_matched,
                           int emit_unmatched_probe, void* stream) {
  if (n_probe == 0) return 0;
  hipLaunchKernelGGL(k_join_fill, dim3(au_grid(n_probe, 256)), dim3(256), 0,
                     (hipStream_t)stream, (const AuColDesc*)bcols,
                     (const AuColDesc*)pcols, ncols, n_probe, heads,
                     (uint32_t)(cap - 1), next, phashes, offsets, build_out,
                     probe_out, build_matched, emit_unmatched_probe);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------- partitioning
// part_id = pmod(murmur3_hash, nparts)  (Spark HashPartitioning semantics)
__global__ void k_pmod(const int32_t* hashes, int64_t n, int32_t nparts,
                       int32_t* part_ids) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t m = hashes[i] % nparts;
    part_ids[i] = m < 0 ? m + nparts : m;
  }
}

AU_EXPORT int au_pmod(const int32_t* hashes, int64_t n, int32_t nparts,
                      int32_t* part_ids, void* stream) {
  if (n == 0) return 0;
  hipLaunchKernelGGL(k_pmod, dim3(au_grid(n, 256)), dim3(256), 0,
                     (hipStream_t)stream, hashes, n, nparts, part_ids);
  return (int)hipGetLastError();
}

// Stable counting-sort scatter by partition id.
// Pass 1: histogram (atomics). Host computes exclusive prefix (torch cumsum).
// Pass 2: stable scatter — each row's final slot = base[part] + rank of row
// within its partition. Rank computed via one atomicAdd per row gives an
// arbitrary intra-partition order; stability is not required by the shuffle
// contract (Spark shuffle reader order is unspecified).
__global__ void k_part_hist(const int32_t* part_ids, int64_t n, int32_t nparts,
                            int32_t* counts) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    atomicAdd(&counts[part_ids[i]], 1);
  }
}

__global__ void k_part_scatter(const int32_t* part_ids, int64_t n,
                               const int64_t* base, int32_t* cursors,
                               int64_t* order) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t p = part_ids[i];
    int64_t slot = base[p] + atomicAdd(&cursors[p], 1);
    order[slot] = i;
  }
}

AU_EXPORT int au_part_hist(const int32_t* part_ids, int64_t n, int32_t nparts,
                           int32_t* counts, void* stream) {
  if (n == 0) return 0;
  hipLaunchKernelGGL(k_part_hist, dim3(au_grid(n, 256)), dim3(256), 0,
                     (hipStream_t)stream, part_ids, n, nparts, counts);
  return (int)hipGetLastError();
}

AU_EXPORT int au_part_scatter(const int32_t* part_ids, int64_t n,
                              const int64_t* base, int32_t* cursors,
                              int64_t* order, void* stream) {
  if (n == 0) return 0;
  hipLaunchKernelGGL(k_part_scatter, dim3(au_grid(n, 256)), dim3(256), 0,
                     (hipStream_t)stream, part_ids, n, base, cursors, order);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------- byte gather
// Variable-length row gather for string columns (arrow/selection.rs take()
// analogue): one wavefront per output row copies its bytes, replacing the
// arange/repeat_interleave/fancy-index chain (5 torch kernels) the r2
// profile showed on every string gather.
__global__ void k_bytes_gather(const uint8_t* __restrict__ src,
                               const int64_t* __restrict__ src_start,
                               const int64_t* __restrict__ out_off,
                               uint8_t* __restrict__ out, int64_t nrows) {
  // thread-per-row: TPC-DS strings average 10-30 bytes, so row-level
  // parallelism beats wavefront-per-row lane masking; long rows chunk
  // through an 8-byte loop
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       row < nrows; row += (int64_t)gridDim.x * blockDim.x) {
    const int64_t dst0 = out_off[row];
    const int64_t len = out_off[row + 1] - dst0;
    const int64_t s0 = src_start[row];
    int64_t j = 0;
    for (; j + 8 <= len; j += 8) {
      uint64_t w;
      __builtin_memcpy(&w, src + s0 + j, 8);
      __builtin_memcpy(out + dst0 + j, &w, 8);
    }
    for (; j < len; j++) out[dst0 + j] = src[s0 + j];
  }
}

AU_EXPORT int au_bytes_gather(const uint8_t* src, const int64_t* src_start,
                              const int64_t* out_off, uint8_t* out,
                              int64_t nrows, void* stream) {
  if (nrows == 0) return 0;
  int64_t blocks = (nrows + 255) / 256;
  if (blocks > 16384) blocks = 16384;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(k_bytes_gather, dim3((uint32_t)blocks), dim3(256), 0,
                     (hipStream_t)stream, src, src_start, out_off, out, nrows);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------------ version
AU_EXPORT int au_abi_version() { return 1; }

// ------------------------------------------------------- fused gather
// One launch gathers ALL fixed-width columns of a batch (join/window/
// sort/partition row gathers were one at::native index_select launch per
// column — the top at::native entry of the final SF=10 profile).
// Negative row index => null output (outer-join semantics).
#define MG_MAX_COLS 24

struct MgCol {
  const void* src;
  const uint8_t* src_valid;  // nullptr = all valid
  void* dst;
  uint8_t* dst_valid;        // nullptr = caller wants no validity
  int64_t esize;             // 1/2/4/8/16
  int64_t pad;
};

struct MgParams { MgCol cols[MG_MAX_COLS]; };

__global__ void __launch_bounds__(256) k_multi_gather(
    const int64_t* __restrict__ idx, int64_t n, MgParams p, int ncols) {
  // blocks tile (col, row-chunk): consecutive blocks cover the same col
  int64_t chunks_per_col = (n + 255) / 256;
  for (int64_t b = blockIdx.x; b < chunks_per_col * ncols; b += gridDim.x) {
    int c = (int)(b / chunks_per_col);
    int64_t row = (b % chunks_per_col) * 256 + threadIdx.x;
    if (row >= n) continue;
    const MgCol col = p.cols[c];
    int64_t i = idx[row];
    bool ok = i >= 0;
    int64_t src_i = ok ? i : 0;
    bool v = ok && (!col.src_valid || col.src_valid[src_i]);
    if (col.dst_valid) col.dst_valid[row] = v;
    switch (col.esize) {
      case 1: ((uint8_t*)col.dst)[row] = ok ? ((const uint8_t*)col.src)[src_i] : 0; break;
      case 2: ((uint16_t*)col.dst)[row] = ok ? ((const uint16_t*)col.src)[src_i] : 0; break;
      case 4: ((uint32_t*)col.dst)[row] = ok ? ((const uint32_t*)col.src)[src_i] : 0; break;
      case 8: ((uint64_t*)col.dst)[row] = ok ? ((const uint64_t*)col.src)[src_i] : 0; break;
      case 16: {
        ulonglong2 z; z.x = 0; z.y = 0;
        ((ulonglong2*)col.dst)[row] = ok ? ((const ulonglong2*)col.src)[src_i] : z;
        break;
      }
      default: break;
    }
  }
}

AU_EXPORT int au_multi_gather(const void* idx, int64_t n,
                              const void* host_cols, int ncols,
                              void* stream) {
  if (n <= 0 || ncols <= 0) return 0;
  if (ncols > MG_MAX_COLS) return 1002;
  MgParams p{};
  const MgCol* hc = (const MgCol*)host_cols;
  for (int i = 0; i < ncols; i++) p.cols[i] = hc[i];
  int64_t chunks = ((n + 255) / 256) * ncols;
  int64_t g = chunks > 4096 ? 4096 : chunks;
  hipLaunchKernelGGL(k_multi_gather, dim3((uint32_t)g), dim3(256), 0,
                     (hipStream_t)stream, (const int64_t*)idx, n, p, ncols);
  return (int)hipGetLastError();
}
