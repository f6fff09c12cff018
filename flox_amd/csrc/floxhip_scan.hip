/* floxhip — grouped scans (cumsum / nancumsum / ffill / bfill).
 *
 * Re-implements the reference's eager scan path (flox/scan.py:101-352
 * groupby_scan -> chunk_scan; flox/aggregate_flox.py:269-325 ffill /
 * _np_grouped_scan): rows are stably sorted by group code so each group is
 * one contiguous run IN ORIGINAL ROW ORDER, a device scan-by-key runs the
 * per-group recurrence, and results scatter back through the permutation.
 *
 * Ops: 0 = cumsum (NaN poisons the rest of its group, like np.cumsum),
 *      1 = nancumsum (NaN contributes 0, np.nancumsum),
 *      2 = ffill, 3 = bfill (carry last/next non-NaN within the group).
 * Rows whose labels fall outside the expected groups form their own
 * trailing group, exactly like the reference's NaN-sentinel group
 * (factorize.py:201-210).
 */

#include <hip/hip_runtime.h>
#include <cstring>
#include <cstdint>
#include <rocprim/rocprim.hpp>

#include "../../include/floxhip.h"

#define FHS_CHECK(x)                             \
  do {                                           \
    hipError_t _e = (x);                         \
    if (_e != hipSuccess) return (int)_e + 1000; \
  } while (0)

namespace {

enum { SCAN_CUMSUM = 0, SCAN_NANCUMSUM = 1, SCAN_FFILL = 2, SCAN_BFILL = 3 };

template <typename V>
struct FillPair {
  V v;
  int32_t valid;
};

template <typename V>
struct FillOp {
  __device__ FillPair<V> operator()(const FillPair<V>& a, const FillPair<V>& b) const {
    return b.valid ? b : a;
  }
};

template <typename V>
__device__ __forceinline__ bool snan(V v) {
  if (std::is_same<V, float>::value || std::is_same<V, double>::value) return v != v;
  return false;
}

template <typename L>
__global__ void k_spack(const L* __restrict__ labels, const L* __restrict__ labels2,
                        int64_t n, int64_t ngroups, int64_t g0, int64_t g1,
                        uint32_t* __restrict__ codes, uint32_t* __restrict__ idx) {
  const bool twolab = labels2 != nullptr;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    int64_t code;
    const int64_t l0 = (int64_t)labels[i];
    if (twolab) {
      const int64_t l1 = (int64_t)labels2[i];
      code = ((uint64_t)l0 >= (uint64_t)g0 || (uint64_t)l1 >= (uint64_t)g1)
                 ? ngroups
                 : l0 * g1 + l1;
    } else {
      code = ((uint64_t)l0 >= (uint64_t)ngroups) ? ngroups : l0;
    }
    codes[i] = (uint32_t)code;
    idx[i] = (uint32_t)i;
  }
}

/* gather values into sorted order, building the scan input; perm == NULL
 * means the labels were already sorted (FH_SORTED_LABELS) and rows are
 * read in place */
template <typename V, int OP>
__global__ void k_sgather(const V* __restrict__ values, const uint32_t* __restrict__ perm,
                          int64_t n, V* __restrict__ sv, FillPair<V>* __restrict__ sp) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    /* bfill scans the reversed sequence */
    const int64_t j = (OP == SCAN_BFILL) ? (n - 1 - i) : i;
    const V v = values[perm ? (int64_t)perm[j] : j];
    if (OP == SCAN_CUMSUM) {
      sv[i] = v;
    } else if (OP == SCAN_NANCUMSUM) {
      sv[i] = snan(v) ? (V)0 : v;
    } else {
      sp[i].v = v;
      sp[i].valid = snan(v) ? 0 : 1;
    }
  }
}

template <int OP>
__global__ void k_skeys_rev(const uint32_t* __restrict__ codes_sorted, int64_t n,
                            uint32_t* __restrict__ rkeys) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    rkeys[i] = codes_sorted[n - 1 - i];
}

/* scatter results back to original row order */
template <typename V, int OP>
__global__ void k_sscatter(const V* __restrict__ sv, const FillPair<V>* __restrict__ sp,
                           const uint32_t* __restrict__ perm, int64_t n,
                           V* __restrict__ out) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const double NAN_ = __longlong_as_double(0x7FF8000000000000ll);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const int64_t j = (OP == SCAN_BFILL) ? (n - 1 - i) : i;
    const int64_t o = perm ? (int64_t)perm[j] : j;
    if (OP == SCAN_CUMSUM || OP == SCAN_NANCUMSUM) {
      out[o] = sv[i];
    } else {
      const FillPair<V> p = sp[i];
      out[o] = p.valid ? p.v : (V)NAN_; /* leading gap stays NaN */
    }
  }
}

template <typename V, typename L, int OP>
int run_scan(fh_call* c, V* out) {
  hipStream_t stream = (hipStream_t)c->stream;
  const int64_t n = c->n;
  char* scr = (char*)c->scratch;
  int64_t o = 0;
  auto carve = [&](int64_t b) {
    int64_t r = o;
    o += ((b + 255) / 256) * 256;
    return r;
  };
  uint32_t* codes = (uint32_t*)(scr + carve(n * 4));
  uint32_t* codes_s = (uint32_t*)(scr + carve(n * 4));
  uint32_t* idx = (uint32_t*)(scr + carve(n * 4));
  uint32_t* perm = (uint32_t*)(scr + carve(n * 4));
  constexpr bool IS_FILL = OP == SCAN_FFILL || OP == SCAN_BFILL;
  V* sv = nullptr;
  V* sv2 = nullptr;
  FillPair<V>* sp = nullptr;
  FillPair<V>* sp2 = nullptr;
  uint32_t* rkeys = nullptr;
  if (IS_FILL) {
    sp = (FillPair<V>*)(scr + carve(n * (int64_t)sizeof(FillPair<V>)));
    sp2 = (FillPair<V>*)(scr + carve(n * (int64_t)sizeof(FillPair<V>)));
  } else {
    sv = (V*)(scr + carve(n * (int64_t)sizeof(V)));
    sv2 = (V*)(scr + carve(n * (int64_t)sizeof(V)));
  }
  if (OP == SCAN_BFILL) rkeys = (uint32_t*)(scr + carve(n * 4));

  size_t ts = 0, tscan = 0;
  (void)rocprim::radix_sort_pairs(nullptr, ts, codes, codes_s, idx, perm, (size_t)n, 0, 32, stream);
  if (IS_FILL)
    (void)rocprim::inclusive_scan_by_key(nullptr, tscan, codes_s, sp, sp2, (size_t)n,
                                   FillOp<V>(), rocprim::equal_to<uint32_t>(), stream);
  else
    (void)rocprim::inclusive_scan_by_key(nullptr, tscan, codes_s, sv, sv2, (size_t)n,
                                   rocprim::plus<V>(), rocprim::equal_to<uint32_t>(), stream);
  void* temp = scr + carve((int64_t)std::max(ts, tscan));
  if (o > c->scratch_bytes) return 3;

  const int grid = (int)std::min<int64_t>((n + 255) / 256, 2048) + 1;
  const bool presorted = (c->flags & FH_SORTED_LABELS) != 0;
  if (presorted) {
    /* caller guarantees nondecreasing in-range labels (the reference's
     * issorted fast path, aggregate_flox.py:9-23): pack the u32 keys but
     * skip the radix sort, gather and scatter in place */
    hipLaunchKernelGGL((k_spack<L>), dim3(grid), dim3(256), 0, stream,
                       (const L*)c->labels, (const L*)c->labels2, n, c->ngroups,
                       c->g0, c->g1, codes_s, idx);
    FHS_CHECK(hipGetLastError());
    perm = nullptr;
  } else {
    hipLaunchKernelGGL((k_spack<L>), dim3(grid), dim3(256), 0, stream,
                       (const L*)c->labels, (const L*)c->labels2, n, c->ngroups,
                       c->g0, c->g1, codes, idx);
    FHS_CHECK(hipGetLastError());
    FHS_CHECK(rocprim::radix_sort_pairs(temp, ts, codes, codes_s, idx, perm,
                                        (size_t)n, 0, 32, stream));
  }
  hipLaunchKernelGGL((k_sgather<V, OP>), dim3(grid), dim3(256), 0, stream,
                     (const V*)c->values, perm, n, sv, sp);
  FHS_CHECK(hipGetLastError());
  const uint32_t* keys = codes_s;
  if (OP == SCAN_BFILL) {
    hipLaunchKernelGGL((k_skeys_rev<OP>), dim3(grid), dim3(256), 0, stream,
                       codes_s, n, rkeys);
    FHS_CHECK(hipGetLastError());
    keys = rkeys;
  }
  if (IS_FILL)
    FHS_CHECK(rocprim::inclusive_scan_by_key(temp, tscan, keys, sp, sp2, (size_t)n,
                                             FillOp<V>(), rocprim::equal_to<uint32_t>(), stream));
  else
    FHS_CHECK(rocprim::inclusive_scan_by_key(temp, tscan, keys, sv, sv2, (size_t)n,
                                             rocprim::plus<V>(), rocprim::equal_to<uint32_t>(), stream));
  hipLaunchKernelGGL((k_sscatter<V, OP>), dim3(grid), dim3(256), 0, stream, sv2,
                     sp2, perm, n, out);
  FHS_CHECK(hipGetLastError());
  return 0;
}

template <typename V, typename L>
int dispatch_scan_op(fh_call* c, int op, V* out) {
  switch (op) {
    case SCAN_CUMSUM: return run_scan<V, L, SCAN_CUMSUM>(c, out);
    case SCAN_NANCUMSUM: return run_scan<V, L, SCAN_NANCUMSUM>(c, out);
    case SCAN_FFILL: return run_scan<V, L, SCAN_FFILL>(c, out);
    case SCAN_BFILL: return run_scan<V, L, SCAN_BFILL>(c, out);
    default: return 4;
  }
}

}  // namespace

extern "C" {

int64_t fh_scan_scratch_bytes(const fh_call* c) {
  const int64_t n = c->n;
  auto al = [](int64_t b) { return ((b + 255) / 256) * 256; };
  const int64_t vsz = (c->vdtype == FH_F64 || c->vdtype == FH_I64) ? 8 : 4;
  const int64_t psz = (c->vdtype == FH_F64 || c->vdtype == FH_I64) ? 16 : 8;
  size_t ts = 0, t2 = 0, t3 = 0;
  (void)rocprim::radix_sort_pairs(nullptr, ts, (const uint32_t*)nullptr, (uint32_t*)nullptr,
                            (const uint32_t*)nullptr, (uint32_t*)nullptr, (size_t)n,
                            0, 32, 0);
  (void)rocprim::inclusive_scan_by_key(nullptr, t2, (const uint32_t*)nullptr,
                                 (const double*)nullptr, (double*)nullptr, (size_t)n,
                                 rocprim::plus<double>(), rocprim::equal_to<uint32_t>(), 0);
  /* the fill ops scan 16-byte pairs, which need more rocprim temp */
  (void)rocprim::inclusive_scan_by_key(nullptr, t3, (const uint32_t*)nullptr,
                                 (const FillPair<double>*)nullptr, (FillPair<double>*)nullptr,
                                 (size_t)n, FillOp<double>(), rocprim::equal_to<uint32_t>(), 0);
  /* 4 x u32 arrays + reversed keys + 2 scan buffers (pair-sized upper bound) */
  return 5 * al(n * 4) + 2 * al(n * psz) +
         al((int64_t)std::max(ts, std::max(t2, t3))) + 2 * al(n * vsz);
}

/* grouped scan; op: 0 cumsum, 1 nancumsum, 2 ffill, 3 bfill.
 * out_sum receives value-dtype[n] in ORIGINAL row order. */
int fh_grouped_scan(fh_call* c, int op) {
  if (!c || !c->values || !c->labels || !c->out_sum) return 6;
  switch (c->vdtype) {
    case FH_F32:
      return c->ldtype == FH_L_I64 ? dispatch_scan_op<float, int64_t>(c, op, (float*)c->out_sum)
                                   : dispatch_scan_op<float, int32_t>(c, op, (float*)c->out_sum);
    case FH_F64:
      return c->ldtype == FH_L_I64 ? dispatch_scan_op<double, int64_t>(c, op, (double*)c->out_sum)
                                   : dispatch_scan_op<double, int32_t>(c, op, (double*)c->out_sum);
    case FH_I64:
      return c->ldtype == FH_L_I64 ? dispatch_scan_op<int64_t, int64_t>(c, op, (int64_t*)c->out_sum)
                                   : dispatch_scan_op<int64_t, int32_t>(c, op, (int64_t*)c->out_sum);
    case FH_I32:
      return c->ldtype == FH_L_I64 ? dispatch_scan_op<int32_t, int64_t>(c, op, (int32_t*)c->out_sum)
                                   : dispatch_scan_op<int32_t, int32_t>(c, op, (int32_t*)c->out_sum);
    default:
      return 9;
  }
}

}  /* extern "C" */
