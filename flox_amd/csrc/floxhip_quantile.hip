/* floxhip — sorted-quantile family (quantile/nanquantile/median/nanmedian).
 *
 * The GPU analogue of the reference's group-aware quantile
 * (flox/aggregate_flox.py:50-130): instead of the complex-number partition
 * trick, rows are radix-sorted by the packed key (group code, encoded
 * value) so every group's values lie ascending in one contiguous run, with
 * NaNs (canonicalized) at the run's tail; a segmented kernel then
 * binary-searches each group's run boundaries + NaN boundary and computes
 * numpy's method="linear" interpolation (reference _lerp,
 * aggregate_flox.py:26-47).
 *
 * f32/i32 values: ONE keys-only 64-bit radix sort (code<<32 | enc32(v)).
 * f64/i64 values: sort by enc64(v) carrying the code, then a stable 32-bit
 * sort by code carrying enc64(v).
 * Sorts are rocprim::radix_sort_{keys,pairs} (device-wide LSD radix).
 */

#include <hip/hip_runtime.h>
#include <cstring>
#include <cstdint>
#include <rocprim/rocprim.hpp>

#include "../../include/floxhip.h"

#define FHQ_CHECK(x)                                \
  do {                                              \
    hipError_t _e = (x);                            \
    if (_e != hipSuccess) return (int)_e + 1000;    \
  } while (0)

namespace {

constexpr uint32_t INVALID_CODE = 0xFFFFFFFFu;

/* order-preserving encodings with a single canonical NaN that sorts last */
__device__ __forceinline__ uint32_t enc32f(float v) {
  if (v != v) return 0xFFFFFFFFu; /* canonical NaN: after every real/inf */
  uint32_t u = __float_as_uint(v);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}
__device__ __forceinline__ float dec32f(uint32_t u) {
  if (u == 0xFFFFFFFFu) return __uint_as_float(0x7FC00000u);
  uint32_t r = (u & 0x80000000u) ? (u ^ 0x80000000u) : ~u;
  return __uint_as_float(r);
}
__device__ __forceinline__ uint32_t enc32i(int32_t v) { return (uint32_t)v ^ 0x80000000u; }
__device__ __forceinline__ int32_t dec32i(uint32_t u) { return (int32_t)(u ^ 0x80000000u); }

__device__ __forceinline__ uint64_t enc64f(double v) {
  if (v != v) return ~0ull;
  uint64_t u = __double_as_longlong(v);
  return (u & 0x8000000000000000ull) ? ~u : (u | 0x8000000000000000ull);
}
__device__ __forceinline__ double dec64f(uint64_t u) {
  if (u == ~0ull) return __longlong_as_double(0x7FF8000000000000ll);
  uint64_t r = (u & 0x8000000000000000ull) ? (u ^ 0x8000000000000000ull) : ~u;
  return __longlong_as_double((long long)r);
}
__device__ __forceinline__ uint64_t enc64i(int64_t v) {
  return (uint64_t)v ^ 0x8000000000000000ull;
}
__device__ __forceinline__ int64_t dec64i(uint64_t u) {
  return (int64_t)(u ^ 0x8000000000000000ull);
}

__device__ __forceinline__ int64_t qcode_of(int64_t l0, int64_t l1, bool twolab,
                                            int64_t g0, int64_t g1, int64_t ngroups) {
  if (twolab) {
    if ((uint64_t)l0 >= (uint64_t)g0 || (uint64_t)l1 >= (uint64_t)g1) return -1;
    return l0 * g1 + l1;
  }
  return ((uint64_t)l0 >= (uint64_t)ngroups) ? -1 : l0;
}

/* pack (code, value) into one u64 key (4-byte value dtypes) */
template <typename V, typename L>
__global__ void k_qpack32(const V* __restrict__ values, const L* __restrict__ labels,
                          const L* __restrict__ labels2, int64_t n, int64_t ngroups,
                          int64_t g0, int64_t g1, uint64_t* __restrict__ keys) {
  const bool twolab = labels2 != nullptr;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const int64_t code = qcode_of((int64_t)labels[i], twolab ? (int64_t)labels2[i] : 0,
                                  twolab, g0, g1, ngroups);
    uint32_t e;
    if (std::is_same<V, float>::value)
      e = enc32f((float)values[i]);
    else
      e = enc32i((int32_t)values[i]);
    keys[i] = (code < 0) ? ~0ull : (((uint64_t)(uint32_t)code) << 32) | e;
  }
}

/* 8-byte value dtypes: encoded value + separate code array */
template <typename V, typename L>
__global__ void k_qpack64(const V* __restrict__ values, const L* __restrict__ labels,
                          const L* __restrict__ labels2, int64_t n, int64_t ngroups,
                          int64_t g0, int64_t g1, uint64_t* __restrict__ enc,
                          uint32_t* __restrict__ codes) {
  const bool twolab = labels2 != nullptr;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const int64_t code = qcode_of((int64_t)labels[i], twolab ? (int64_t)labels2[i] : 0,
                                  twolab, g0, g1, ngroups);
    if (std::is_same<V, double>::value)
      enc[i] = enc64f((double)values[i]);
    else
      enc[i] = enc64i((int64_t)values[i]);
    codes[i] = (code < 0) ? INVALID_CODE : (uint32_t)code;
  }
}

/* per-group run boundaries by binary search over the sorted codes */
template <bool PACKED>
__global__ void k_qoffsets(const uint64_t* __restrict__ keys,
                           const uint32_t* __restrict__ codes, int64_t n,
                           int64_t ngroups, int64_t* __restrict__ off) {
  const int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (g > ngroups) return;
  int64_t lo = 0, hi = n;
  while (lo < hi) {
    const int64_t mid = (lo + hi) >> 1;
    const uint32_t c = PACKED ? (uint32_t)(keys[mid] >> 32) : codes[mid];
    if ((int64_t)c < g)
      lo = mid + 1;
    else
      hi = mid;
  }
  off[g] = lo;
}

/* numpy method="linear" quantile per (group, q); reference _lerp
 * (aggregate_flox.py:26-47) including the t>=0.5 form */
template <typename V, bool PACKED>
__global__ void k_quantile(const uint64_t* __restrict__ keys, int64_t n,
                           const int64_t* __restrict__ off, int64_t ngroups,
                           const double* __restrict__ q, int nq, int skipna,
                           double* __restrict__ out) {
  const int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ngroups * nq) return;
  const int64_t g = t % ngroups;
  const int qi = (int)(t / ngroups);
  const int64_t start = off[g], end = off[g + 1];
  const int64_t full = end - start;
  const double NAN_ = __longlong_as_double(0x7FF8000000000000ll);
  double res = NAN_;
  const bool is_f = std::is_same<V, float>::value || std::is_same<V, double>::value;
  if (full > 0) {
    int64_t nan_start = end;
    if (is_f) {
      /* canonical NaNs sort last within the run */
      int64_t lo = start, hi = end;
      const uint64_t nan_key = PACKED ? ((((uint64_t)(uint32_t)g) << 32) | 0xFFFFFFFFu) : ~0ull;
      while (lo < hi) {
        const int64_t mid = (lo + hi) >> 1;
        if (keys[mid] < nan_key)
          lo = mid + 1;
        else
          hi = mid;
      }
      nan_start = lo;
    }
    const int64_t actual = nan_start - start;
    const bool nanmask = actual != full;
    if (actual > 0 && (skipna || !nanmask)) {
      const double vi = q[qi] * (double)(actual - 1);
      int64_t lo_i = (int64_t)floor(vi), hi_i = (int64_t)ceil(vi);
      if (lo_i < 0) lo_i = 0;
      if (hi_i > actual - 1) hi_i = actual - 1;
      auto val_at = [&](int64_t k) -> double {
        const uint64_t key = keys[start + k];
        if (PACKED) {
          const uint32_t e = (uint32_t)key;
          if (std::is_same<V, float>::value) return (double)dec32f(e);
          return (double)dec32i(e);
        }
        if (std::is_same<V, double>::value) return dec64f(key);
        return (double)dec64i(key);
      };
      const double a = val_at(lo_i), b = val_at(hi_i);
      const double tq = vi - (double)lo_i;
      const double diff = b - a;
      res = (tq >= 0.5) ? (b - diff * (1.0 - tq)) : (a + diff * tq);
    }
  }
  out[(int64_t)qi * ngroups + g] = res;
}

/* mode: longest equal-value run in each group's sorted span; sorted order
 * makes scipy's tie rule (smallest of the modes) the first-found run.
 * nan_policy: "propagate" (mode) -> NaN result when the group has NaNs;
 * "omit" (nanmode) -> NaN tail ignored (reference aggregate_npg.py:185-215). */
template <typename V, bool PACKED>
__global__ void k_mode(const uint64_t* __restrict__ keys, int64_t n,
                       const int64_t* __restrict__ off, int64_t ngroups,
                       int skipna, V* __restrict__ out) {
  const int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= ngroups) return;
  const int64_t start = off[g], end = off[g + 1];
  const bool is_f = std::is_same<V, float>::value || std::is_same<V, double>::value;
  const double NAN_ = __longlong_as_double(0x7FF8000000000000ll);
  auto write_nan_or_zero = [&]() { out[g] = is_f ? (V)NAN_ : (V)0; };
  if (end <= start) {
    write_nan_or_zero();
    return;
  }
  int64_t nan_start = end;
  if (is_f) {
    int64_t lo = start, hi = end;
    const uint64_t nan_key = PACKED ? ((((uint64_t)(uint32_t)g) << 32) | 0xFFFFFFFFu) : ~0ull;
    while (lo < hi) {
      const int64_t mid = (lo + hi) >> 1;
      if (keys[mid] < nan_key)
        lo = mid + 1;
      else
        hi = mid;
    }
    nan_start = lo;
  }
  if (nan_start == start) {
    /* no non-NaN values: NaN either way (scipy omit warns and returns NaN;
     * propagate's most-common value IS NaN) */
    write_nan_or_zero();
    return;
  }
  uint64_t best = keys[start], cur = keys[start];
  int64_t best_n = 1, cur_n = 1;
  for (int64_t i = start + 1; i < nan_start; ++i) {
    const uint64_t k = keys[i];
    if (k == cur) {
      ++cur_n;
    } else {
      if (cur_n > best_n) {
        best = cur;
        best_n = cur_n;
      }
      cur = k;
      cur_n = 1;
    }
  }
  if (cur_n > best_n) {
    best = cur;
    best_n = cur_n;
  }
  if (!skipna && (end - nan_start) > best_n) {
    /* scipy propagate counts NaNs as a value: a strictly-most-frequent NaN
     * run wins (ties resolve to the smallest, i.e. any non-NaN) */
    write_nan_or_zero();
    return;
  }
  if (PACKED) {
    const uint32_t e = (uint32_t)best;
    if (std::is_same<V, float>::value)
      out[g] = (V)dec32f(e);
    else
      out[g] = (V)dec32i((int32_t)e);
  } else {
    if (std::is_same<V, double>::value)
      out[g] = (V)dec64f(best);
    else
      out[g] = (V)dec64i(best);
  }
}

template <typename V, typename L>
int run_quantile(fh_call* c, const double* q_dev, int nq, double* out) {
  hipStream_t stream = (hipStream_t)c->stream;
  const int skipna = (c->flags & FH_SKIPNAN) ? 1 : 0;
  const int64_t n = c->n, ngroups = c->ngroups;
  char* scr = (char*)c->scratch;
  constexpr bool PACKED = sizeof(V) == 4;

  int64_t off0 = 0;
  auto carve = [&](int64_t b) {
    int64_t o = off0;
    off0 += ((b + 255) / 256) * 256;
    return o;
  };

  const int grid = (int)std::min<int64_t>((n + 255) / 256, 2048) + 1;
  if (PACKED) {
    uint64_t* kin = (uint64_t*)(scr + carve(n * 8));
    uint64_t* kout = (uint64_t*)(scr + carve(n * 8));
    int64_t* off = (int64_t*)(scr + carve((ngroups + 1) * 8));
    size_t temp_bytes = 0;
    (void)rocprim::radix_sort_keys(nullptr, temp_bytes, kin, kout, (size_t)n, 0, 64, stream);
    void* temp = scr + carve((int64_t)temp_bytes);
    if (off0 > c->scratch_bytes) return 3;
    hipLaunchKernelGGL((k_qpack32<V, L>), dim3(grid), dim3(256), 0, stream,
                       (const V*)c->values, (const L*)c->labels,
                       (const L*)c->labels2, n, ngroups, c->g0, c->g1, kin);
    FHQ_CHECK(hipGetLastError());
    FHQ_CHECK(rocprim::radix_sort_keys(temp, temp_bytes, kin, kout, (size_t)n, 0, 64, stream));
    int ob = (int)((ngroups + 1 + 255) / 256);
    hipLaunchKernelGGL((k_qoffsets<true>), dim3(ob), dim3(256), 0, stream, kout,
                       (const uint32_t*)nullptr, n, ngroups, off);
    FHQ_CHECK(hipGetLastError());
    if (nq == 0) { /* mode */
      int mb = (int)((ngroups + 255) / 256);
      hipLaunchKernelGGL((k_mode<V, true>), dim3(mb), dim3(256), 0, stream,
                         kout, n, off, ngroups, skipna, (V*)out);
      FHQ_CHECK(hipGetLastError());
      return 0;
    }
    int qb = (int)((ngroups * nq + 255) / 256);
    hipLaunchKernelGGL((k_quantile<V, true>), dim3(qb), dim3(256), 0, stream,
                       kout, n, off, ngroups, q_dev, nq, skipna, out);
    FHQ_CHECK(hipGetLastError());
    return 0;
  }

  uint64_t* e_in = (uint64_t*)(scr + carve(n * 8));
  uint64_t* e_out = (uint64_t*)(scr + carve(n * 8));
  uint32_t* c_in = (uint32_t*)(scr + carve(n * 4));
  uint32_t* c_out = (uint32_t*)(scr + carve(n * 4));
  int64_t* off = (int64_t*)(scr + carve((ngroups + 1) * 8));
  size_t t1 = 0, t2 = 0;
  (void)rocprim::radix_sort_pairs(nullptr, t1, e_in, e_out, c_in, c_out, (size_t)n, 0, 64, stream);
  (void)rocprim::radix_sort_pairs(nullptr, t2, c_out, c_in, e_out, e_in, (size_t)n, 0, 32, stream);
  void* temp = scr + carve((int64_t)std::max(t1, t2));
  if (off0 > c->scratch_bytes) return 3;
  hipLaunchKernelGGL((k_qpack64<V, L>), dim3(grid), dim3(256), 0, stream,
                     (const V*)c->values, (const L*)c->labels,
                     (const L*)c->labels2, n, ngroups, c->g0, c->g1, e_in, c_in);
  FHQ_CHECK(hipGetLastError());
  /* sort by encoded value, then stably by code: values ascend within runs */
  FHQ_CHECK(rocprim::radix_sort_pairs(temp, t1, e_in, e_out, c_in, c_out, (size_t)n, 0, 64, stream));
  FHQ_CHECK(rocprim::radix_sort_pairs(temp, t2, c_out, c_in, e_out, e_in, (size_t)n, 0, 32, stream));
  /* sorted codes now in c_in, matching encoded values in e_in */
  int ob = (int)((ngroups + 1 + 255) / 256);
  hipLaunchKernelGGL((k_qoffsets<false>), dim3(ob), dim3(256), 0, stream, e_in,
                     c_in, n, ngroups, off);
  FHQ_CHECK(hipGetLastError());
  if (nq == 0) { /* mode */
    int mb = (int)((ngroups + 255) / 256);
    hipLaunchKernelGGL((k_mode<V, false>), dim3(mb), dim3(256), 0, stream,
                       e_in, n, off, ngroups, skipna, (V*)out);
    FHQ_CHECK(hipGetLastError());
    return 0;
  }
  int qb = (int)((ngroups * nq + 255) / 256);
  hipLaunchKernelGGL((k_quantile<V, false>), dim3(qb), dim3(256), 0, stream,
                     e_in, n, off, ngroups, q_dev, nq, skipna, out);
  FHQ_CHECK(hipGetLastError());
  return 0;
}

}  // namespace

extern "C" {

int64_t fh_quantile_scratch_bytes(const fh_call* c) {
  const int64_t n = c->n, ngroups = c->ngroups;
  size_t temp = 0;
  int64_t bytes = 0;
  auto al = [](int64_t b) { return ((b + 255) / 256) * 256; };
  if (c->vdtype == FH_F32 || c->vdtype == FH_I32) {
    (void)rocprim::radix_sort_keys(nullptr, temp, (const uint64_t*)nullptr,
                             (uint64_t*)nullptr, (size_t)n, 0, 64, 0);
    bytes = 2 * al(n * 8) + al((ngroups + 1) * 8) + al((int64_t)temp);
  } else {
    size_t t1 = 0, t2 = 0;
    (void)rocprim::radix_sort_pairs(nullptr, t1, (const uint64_t*)nullptr,
                              (uint64_t*)nullptr, (const uint32_t*)nullptr,
                              (uint32_t*)nullptr, (size_t)n, 0, 64, 0);
    (void)rocprim::radix_sort_pairs(nullptr, t2, (const uint32_t*)nullptr,
                              (uint32_t*)nullptr, (const uint64_t*)nullptr,
                              (uint64_t*)nullptr, (size_t)n, 0, 32, 0);
    bytes = 2 * al(n * 8) + 2 * al(n * 4) + al((ngroups + 1) * 8) +
            al((int64_t)std::max(t1, t2));
  }
  return bytes;
}

/* grouped quantiles: `means` carries the device q array (f64[nq], smuggled
 * through the existing struct field), g0 (when labels2 is NULL) is unused,
 * out_sum receives f64[nq*ngroups] results (NaN for empty groups, all-NaN
 * groups under skipna, and NaN-containing groups without skipna) */
int fh_grouped_quantile(fh_call* c, int nq) {
  /* nq == 0: grouped mode — out_sum then holds value-dtype[ngroups] */
  if (!c || !c->values || !c->labels || !c->out_sum) return 6;
  if (nq > 0 && !c->means) return 6;
  const double* q_dev = c->means;
  double* out = (double*)c->out_sum;
  switch (c->vdtype) {
    case FH_F32:
      return c->ldtype == FH_L_I64 ? run_quantile<float, int64_t>(c, q_dev, nq, out)
                                   : run_quantile<float, int32_t>(c, q_dev, nq, out);
    case FH_F64:
      return c->ldtype == FH_L_I64 ? run_quantile<double, int64_t>(c, q_dev, nq, out)
                                   : run_quantile<double, int32_t>(c, q_dev, nq, out);
    case FH_I64:
      return c->ldtype == FH_L_I64 ? run_quantile<int64_t, int64_t>(c, q_dev, nq, out)
                                   : run_quantile<int64_t, int32_t>(c, q_dev, nq, out);
    case FH_I32:
      return c->ldtype == FH_L_I64 ? run_quantile<int32_t, int64_t>(c, q_dev, nq, out)
                                   : run_quantile<int32_t, int32_t>(c, q_dev, nq, out);
    default:
      return 9;
  }
}

}  /* extern "C" */
