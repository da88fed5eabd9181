// Aggregate accumulation kernels — gfx950.
//
// Role parity: the reference's per-function accumulators
// (datafusion-ext-plans/src/agg/acc.rs + sum/maxmin/count). On MI355X
// the accumulate is a scatter with device-scope atomics over the
// group-id array produced by k_group_insert (kernels.hip). Replaces
// torch's scatter_add_ path, which measured ~400x slower than the
// atomic scatter on fp64 (rocprof: 5.5s per 14M-row scatter).
//
// value dtypes: f64 (code 0), i64 (code 1), i32 (code 2), f32 (code 3).
// ops: sum=0, min=1, max=2. Accumulators are f64 or i64 (widened).
// count (valid-row count per group) accumulates alongside when
// `counts` is non-null.

#include <hip/hip_runtime.h>
#include <stdint.h>

#define AU_EXPORT extern "C" __attribute__((visibility("default")))

static inline int agg_grid(int64_t n) {
  int64_t g = (n + 255) / 256;
  if (g > 2048) g = 2048;
  if (g < 1) g = 1;
  return (int)g;
}

__device__ __forceinline__ void atomic_min_f64(double* addr, double v) {
  unsigned long long* a = (unsigned long long*)addr;
  unsigned long long old = *a, assumed;
  do {
    assumed = old;
    double cur = __longlong_as_double((long long)assumed);
    if (v >= cur) return;
    old = atomicCAS(a, assumed, (unsigned long long)__double_as_longlong(v));
  } while (old != assumed);
}

__device__ __forceinline__ void atomic_max_f64(double* addr, double v) {
  unsigned long long* a = (unsigned long long*)addr;
  unsigned long long old = *a, assumed;
  do {
    assumed = old;
    double cur = __longlong_as_double((long long)assumed);
    if (v <= cur) return;
    old = atomicCAS(a, assumed, (unsigned long long)__double_as_longlong(v));
  } while (old != assumed);
}

__device__ __forceinline__ void atomic_min_i64(int64_t* addr, int64_t v) {
  unsigned long long* a = (unsigned long long*)addr;
  unsigned long long old = *a, assumed;
  do {
    assumed = old;
    if (v >= (int64_t)assumed) return;
    old = atomicCAS(a, assumed, (unsigned long long)v);
  } while (old != assumed);
}

__device__ __forceinline__ void atomic_max_i64(int64_t* addr, int64_t v) {
  unsigned long long* a = (unsigned long long*)addr;
  unsigned long long old = *a, assumed;
  do {
    assumed = old;
    if (v <= (int64_t)assumed) return;
    old = atomicCAS(a, assumed, (unsigned long long)v);
  } while (old != assumed);
}

template <typename VT, bool ACC_F64>
__global__ void k_agg_scatter(const int64_t* gids, int64_t n, const uint8_t* validity,
                              const VT* values, int op, void* acc, int64_t* counts) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (validity && !validity[i]) continue;
    int64_t g = gids[i];
    if (counts) atomicAdd((unsigned long long*)&counts[g], 1ull);
    if (ACC_F64) {
      double v = (double)values[i];
      double* a = (double*)acc;
      if (op == 0) atomicAdd(&a[g], v);
      else if (op == 1) atomic_min_f64(&a[g], v);
      else atomic_max_f64(&a[g], v);
    } else {
      int64_t v = (int64_t)values[i];
      int64_t* a = (int64_t*)acc;
      if (op == 0) atomicAdd((unsigned long long*)&a[g], (unsigned long long)v);
      else if (op == 1) atomic_min_i64(&a[g], v);
      else atomic_max_i64(&a[g], v);
    }
  }
}

// count-only (count_star / count of a validity-masked column)
__global__ void k_agg_count(const int64_t* gids, int64_t n, const uint8_t* validity,
                            int64_t* counts) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (validity && !validity[i]) continue;
    atomicAdd((unsigned long long*)&counts[gids[i]], 1ull);
  }
}

AU_EXPORT int au_agg_scatter(const int64_t* gids, int64_t n, const uint8_t* validity,
                             const void* values, int vtype, int op, int acc_f64,
                             void* acc, int64_t* counts, void* stream) {
  if (n == 0) return 0;
  hipStream_t s = (hipStream_t)stream;
  dim3 g(agg_grid(n)), b(256);
#define CASE(VT)                                                                  \
  if (acc_f64)                                                                    \
    hipLaunchKernelGGL((k_agg_scatter<VT, true>), g, b, 0, s, gids, n, validity,  \
                       (const VT*)values, op, acc, counts);                       \
  else                                                                            \
    hipLaunchKernelGGL((k_agg_scatter<VT, false>), g, b, 0, s, gids, n, validity, \
                       (const VT*)values, op, acc, counts);
  if (vtype == 0) { CASE(double) }
  else if (vtype == 1) { CASE(int64_t) }
  else if (vtype == 2) { CASE(int32_t) }
  else { CASE(float) }
#undef CASE
  return (int)hipGetLastError();
}

AU_EXPORT int au_agg_count(const int64_t* gids, int64_t n, const uint8_t* validity,
                           int64_t* counts, void* stream) {
  if (n == 0) return 0;
  hipLaunchKernelGGL(k_agg_count, dim3(agg_grid(n)), dim3(256), 0,
                     (hipStream_t)stream, gids, n, validity, counts);
  return (int)hipGetLastError();
}
