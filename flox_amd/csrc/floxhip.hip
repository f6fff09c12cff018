/* floxhip — MI355X (gfx950, CDNA4) grouped-reduction kernels.
 *
 * Re-implements, from scratch and HBM-first, the hot path of
 * xarray-contrib/flox: factorize + per-group sum/count/min/max/var
 * (reference flox/core.py:214-394 chunk_reduce, flox/aggregate_flox.py,
 * flox/aggregate_npg.py). Design notes in /root/repo/DESIGN.md.
 *
 * Two paths, selected by group cardinality (the GPU analogue of the
 * reference's _choose_engine, flox/core.py:712-736):
 *
 *  1. LDS-binned scatter (ngroups small enough that the per-group partial
 *     bins fit in the CU's 160 KiB LDS): each 1024-thread workgroup keeps
 *     private bins in LDS, streams its grid-stride share of the rows with
 *     16-byte vector loads (values) + int64 label loads, updates bins with
 *     native LDS atomics (ds_add_f64 / ds_add_u32 / ds_min_u32|u64), then
 *     writes its bins to a per-block slab in HBM with plain coalesced
 *     stores. A combine kernel folds the slab over blocks (fixed order).
 *     Nothing in this path uses inter-workgroup communication.
 *
 *  2. Global-atomic scatter (large ngroups, e.g. the 1e7-group config):
 *     bins live in HBM (LLC-cached), rows update them with device-scope
 *     atomics (global_atomic_add_f64 / _umin / _umax ...). min/max bins are
 *     kept in an order-preserving unsigned encoding so init is a memset and
 *     the atomic is an integer min/max; a decode kernel maps back.
 *
 * Numerics: float32 sums/counts/means/var accumulate in float64 (the
 * numpy_groupies contract, reference tests/test_properties.py:146-151).
 * count/min/max are bit-exact. fp sums use atomics, so the reduction order
 * is not fixed: results are reproducible only to fp-roundoff tolerance
 * (stated in DESIGN.md and the parity tests).
 */

#include <hip/hip_runtime.h>
#include <cstdint>
#include <type_traits>
#include <cstdio>
#include <cstdlib>

#include "../../include/floxhip.h"

#define FH_CHECK(x)                       \
  do {                                    \
    hipError_t _e = (x);                  \
    if (_e != hipSuccess) return (int)_e + 1000; \
  } while (0)

namespace {

constexpr int BLOCK_LDS = 1024;   /* 16 waves/CU at 1 block/CU */
constexpr int BLOCK_ATOMIC = 256;
constexpr int NUM_CU = 256;
constexpr int64_t LDS_MAX = 160 * 1024 - 512; /* gfx950 LDS per workgroup */

/* ---- op-set membership ------------------------------------------------- */
enum {
  B_SUM = 1,
  B_CNT = 2,
  B_PRESENT = 4,
  B_MIN = 8,
  B_MAX = 16,
  B_NANFLAG = 32,
  B_SSD = 64,
  B_PROD = 128,
  B_IDXMIN = 256,
  B_IDXMAX = 512,
  B_WELFORD = 1024,
  B_ARGROW = 2048, /* pair-payload arg-reductions: second bucket pass takes
                      the min row among rows matching the group extremum */
};

__host__ __device__ constexpr int set_bits(int op_set) {
  switch (op_set) {
    case FH_SET_SUM_COUNT: return B_SUM | B_CNT;
    case FH_SET_SUM_COUNT_PRESENT: return B_SUM | B_CNT | B_PRESENT;
    case FH_SET_COUNT: return B_CNT;
    case FH_SET_MIN_FULL: return B_MIN | B_CNT | B_PRESENT | B_NANFLAG;
    case FH_SET_MIN_COUNT: return B_MIN | B_CNT;
    case FH_SET_MAX_FULL: return B_MAX | B_CNT | B_PRESENT | B_NANFLAG;
    case FH_SET_MAX_COUNT: return B_MAX | B_CNT;
    case FH_SET_SSD: return B_SSD;
    case FH_SET_PROD: return B_PROD | B_CNT | B_PRESENT;
    case FH_SET_IDXMIN: return B_IDXMIN | B_CNT | B_PRESENT;
    case FH_SET_IDXMAX: return B_IDXMAX | B_CNT | B_PRESENT;
    case FH_SET_WELFORD: return B_WELFORD | B_CNT;
    case FH_SET_ARGMIN_PAIR: return B_MIN | B_CNT | B_PRESENT | B_NANFLAG | B_ARGROW;
    case FH_SET_ARGMAX_PAIR: return B_MAX | B_CNT | B_PRESENT | B_NANFLAG | B_ARGROW;
    default: return 0;
  }
}

/* ---- per-dtype traits --------------------------------------------------- */
template <typename V> struct Traits;

template <> struct Traits<float> {
  using Acc = double;          /* f64 accumulation (npg contract) */
  using Enc = uint32_t;        /* order-preserving encoding for min/max */
  static constexpr int VEC = 4;
  static __device__ __forceinline__ bool isnan_(float v) { return v != v; }
  static __device__ __forceinline__ Enc enc(float v) {
    uint32_t u = __float_as_uint(v);
    return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
  }
  static __device__ __forceinline__ float dec(Enc u) {
    uint32_t r = (u & 0x80000000u) ? (u ^ 0x80000000u) : ~u;
    return __uint_as_float(r);
  }
  static __device__ __forceinline__ float pos_inf() { return __builtin_inff(); }
};

template <> struct Traits<double> {
  using Acc = double;
  using Enc = uint64_t;
  static constexpr int VEC = 2;
  static __device__ __forceinline__ bool isnan_(double v) { return v != v; }
  static __device__ __forceinline__ Enc enc(double v) {
    uint64_t u = __double_as_longlong(v);
    return (u & 0x8000000000000000ull) ? ~u : (u | 0x8000000000000000ull);
  }
  static __device__ __forceinline__ double dec(Enc u) {
    uint64_t r = (u & 0x8000000000000000ull) ? (u ^ 0x8000000000000000ull) : ~u;
    return __longlong_as_double((long long)r);
  }
  static __device__ __forceinline__ double pos_inf() { return __builtin_inf(); }
};

template <> struct Traits<int32_t> {
  using Acc = int64_t;         /* numpy promotes i32 sums to platform int */
  using Enc = uint32_t;
  static constexpr int VEC = 4;
  static __device__ __forceinline__ bool isnan_(int32_t) { return false; }
  static __device__ __forceinline__ Enc enc(int32_t v) {
    return (uint32_t)v ^ 0x80000000u;   /* order-preserving signed->unsigned */
  }
  static __device__ __forceinline__ int32_t dec(Enc u) { return (int32_t)(u ^ 0x80000000u); }
  static __device__ __forceinline__ int32_t pos_inf() { return INT32_MAX; }
};

template <> struct Traits<int64_t> {
  using Acc = int64_t;
  using Enc = uint64_t;
  static constexpr int VEC = 2;
  static __device__ __forceinline__ bool isnan_(int64_t) { return false; }
  static __device__ __forceinline__ Enc enc(int64_t v) {
    return (uint64_t)v ^ 0x8000000000000000ull;
  }
  static __device__ __forceinline__ int64_t dec(Enc u) {
    return (int64_t)(u ^ 0x8000000000000000ull);
  }
  static __device__ __forceinline__ int64_t pos_inf() { return INT64_MAX; }
};

/* atomic add on the accumulator type (LDS or global pointer) */
__device__ __forceinline__ void acc_add(double* p, double v) { atomicAdd(p, v); }
__device__ __forceinline__ void acc_add(int64_t* p, int64_t v) {
  atomicAdd(reinterpret_cast<unsigned long long*>(p), (unsigned long long)v);
}
__device__ __forceinline__ void enc_min(uint32_t* p, uint32_t v) { atomicMin(p, v); }
__device__ __forceinline__ void enc_min(uint64_t* p, uint64_t v) {
  atomicMin(reinterpret_cast<unsigned long long*>(p), (unsigned long long)v);
}
__device__ __forceinline__ void enc_max(uint32_t* p, uint32_t v) { atomicMax(p, v); }
__device__ __forceinline__ void enc_max(uint64_t* p, uint64_t v) {
  atomicMax(reinterpret_cast<unsigned long long*>(p), (unsigned long long)v);
}
__device__ __forceinline__ void idx_min(int64_t* p, int64_t v) {
  atomicMin(reinterpret_cast<long long*>(p), (long long)v);
}
__device__ __forceinline__ void idx_max(int64_t* p, int64_t v) {
  atomicMax(reinterpret_cast<long long*>(p), (long long)v);
}

/* CAS product (prod is rare; contention-tolerant CAS loop) */
__device__ __forceinline__ void acc_mul(double* p, double v) {
  unsigned long long* u = reinterpret_cast<unsigned long long*>(p);
  unsigned long long old = *u, assumed;
  do {
    assumed = old;
    double cur = __longlong_as_double((long long)assumed);
    old = atomicCAS(u, assumed, (unsigned long long)__double_as_longlong(cur * v));
  } while (old != assumed);
}
__device__ __forceinline__ void acc_mul(int64_t* p, int64_t v) {
  unsigned long long* u = reinterpret_cast<unsigned long long*>(p);
  unsigned long long old = *u, assumed;
  do {
    assumed = old;
    old = atomicCAS(u, assumed, (unsigned long long)((int64_t)assumed * v));
  } while (old != assumed);
}

/* 16-byte-aligned vector type for coalesced loads */
template <typename T, int N> struct alignas(sizeof(T) * N) Vec { T v[N]; };

/* ---- bin layout (shared between LDS carve, slab and combine) ------------ */
struct BinLayout {
  int64_t sum_off, cnt_off, present_off, minmax_off, nanflag_off, sumx_off;
  int64_t bytes;  /* per block-copy of the bins */
};

template <typename V>
__host__ __device__ BinLayout bin_layout(int bits, int64_t ngroups, int64_t cnt_elem_size) {
  BinLayout L{};
  int64_t off = 0;
  auto carve = [&](int64_t elem) {
    int64_t o = off;
    off += ((ngroups * elem + 255) / 256) * 256; /* 256-B aligned sections */
    return o;
  };
  L.sum_off = (bits & (B_SUM | B_SSD | B_PROD | B_IDXMIN | B_IDXMAX | B_WELFORD)) ? carve(8) : -1;
  L.cnt_off = (bits & B_CNT) ? carve(cnt_elem_size) : -1;
  L.present_off = (bits & B_PRESENT) ? carve(4) : -1;
  L.minmax_off = (bits & (B_MIN | B_MAX)) ? carve(sizeof(typename Traits<V>::Enc)) : -1;
  L.nanflag_off = (bits & B_NANFLAG) ? carve(4) : -1;
  L.sumx_off = (bits & B_WELFORD) ? carve(8) : -1;
  L.bytes = off;
  return L;
}

/* ---- kernel 1: LDS-binned grouped reduce -------------------------------- */
template <typename V, typename L, int OPS>
__launch_bounds__(BLOCK_LDS) __global__ void k_reduce_lds(
    const V* __restrict__ values, const L* __restrict__ labels,
    const L* __restrict__ labels2, int64_t n, int64_t ngroups, int64_t g0,
    int64_t g1, const double* __restrict__ means,
    const V* __restrict__ target, int64_t row_offset, int skipnan,
    char* __restrict__ slab, BinLayout lay) {
  using TR = Traits<V>;
  using Acc = typename TR::Acc;
  using SumT = typename std::conditional<
      (OPS & (B_SSD | B_WELFORD)) != 0, double,
      typename std::conditional<(OPS & (B_IDXMIN | B_IDXMAX)) != 0, int64_t, Acc>::type>::type;
  using Enc = typename TR::Enc;
  constexpr int VEC = TR::VEC;
  constexpr bool IS_PROD = (OPS & B_PROD) != 0;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  SumT* s_sum = (OPS & (B_SUM | B_SSD | B_PROD | B_IDXMIN | B_IDXMAX)) ? (SumT*)(smem + lay.sum_off) : nullptr;
  uint32_t* s_cnt = (OPS & B_CNT) ? (uint32_t*)(smem + lay.cnt_off) : nullptr;
  uint32_t* s_present = (OPS & B_PRESENT) ? (uint32_t*)(smem + lay.present_off) : nullptr;
  Enc* s_mm = (OPS & (B_MIN | B_MAX)) ? (Enc*)(smem + lay.minmax_off) : nullptr;
  uint32_t* s_nanflag = (OPS & B_NANFLAG) ? (uint32_t*)(smem + lay.nanflag_off) : nullptr;

  const int tid = threadIdx.x;
  for (int64_t g = tid; g < ngroups; g += blockDim.x) {
    if (OPS & (B_SUM | B_SSD)) s_sum[g] = (SumT)0;
    if (IS_PROD) s_sum[g] = (SumT)1;
    if (OPS & B_IDXMIN) s_sum[g] = (SumT)INT64_MAX;
    if (OPS & B_IDXMAX) s_sum[g] = (SumT)(-1);
    if (OPS & B_CNT) s_cnt[g] = 0u;
    if (OPS & B_PRESENT) s_present[g] = 0u;
    if (OPS & B_MIN) s_mm[g] = (Enc)~(Enc)0;
    if (OPS & B_MAX) s_mm[g] = (Enc)0;
    if (OPS & B_NANFLAG) s_nanflag[g] = 0u;
  }
  __syncthreads();

  const bool twolab = labels2 != nullptr;

  auto process = [&](V v, int64_t l0raw, int64_t l1raw, int64_t row) {
    uint64_t code;
    if (twolab) {
      if ((uint64_t)l0raw >= (uint64_t)g0 || (uint64_t)l1raw >= (uint64_t)g1) return;
      code = (uint64_t)l0raw * (uint64_t)g1 + (uint64_t)l1raw;
    } else {
      if ((uint64_t)l0raw >= (uint64_t)ngroups) return;
      code = (uint64_t)l0raw;
    }
    const bool vnan = TR::isnan_(v);
    if (OPS & B_PRESENT) s_present[code] = 1u;  /* benign write race: all store 1 */
    if (vnan && skipnan) return;
    if (OPS & B_SUM) acc_add(&s_sum[code], (Acc)v);
    if (IS_PROD) acc_mul(&s_sum[code], (Acc)v);
    if (OPS & B_SSD) {
      double d = (double)v - means[code];
      atomicAdd((double*)s_sum + code, d * d);
    }
    if (OPS & (B_IDXMIN | B_IDXMAX)) {
      /* candidate: every surviving row, or rows matching the per-group
       * target value (NaN target matches NaN rows: argmax over data with
       * NaNs lands on the first NaN, as np.argmax does) */
      bool match = true;
      if (target) {
        const V t = target[code];
        match = vnan ? TR::isnan_(t) : (!TR::isnan_(t) && v == t);
      }
      if (match) {
        if (OPS & B_IDXMIN) idx_min((int64_t*)s_sum + code, row + row_offset);
        if (OPS & B_IDXMAX) idx_max((int64_t*)s_sum + code, row + row_offset);
      }
    }
    if (OPS & B_CNT) {
      if (!vnan) atomicAdd(&s_cnt[code], 1u);
    }
    if (OPS & (B_MIN | B_MAX)) {
      if (vnan) {
        if (OPS & B_NANFLAG) s_nanflag[code] = 1u;
      } else {
        if (OPS & B_MIN) enc_min(&s_mm[code], TR::enc(v));
        if (OPS & B_MAX) enc_max(&s_mm[code], TR::enc(v));
      }
    }
  };

  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int64_t gtid = (int64_t)blockIdx.x * blockDim.x + tid;
  const int64_t nvec = n / VEC;
  for (int64_t i = gtid; i < nvec; i += stride) {
    Vec<V, VEC> vv = *reinterpret_cast<const Vec<V, VEC>*>(values + i * VEC);
    Vec<L, VEC> lv = *reinterpret_cast<const Vec<L, VEC>*>(labels + i * VEC);
    if (twolab) {
      Vec<L, VEC> lv2 = *reinterpret_cast<const Vec<L, VEC>*>(labels2 + i * VEC);
#pragma unroll
      for (int k = 0; k < VEC; ++k)
        process(vv.v[k], (int64_t)lv.v[k], (int64_t)lv2.v[k], i * VEC + k);
    } else {
#pragma unroll
      for (int k = 0; k < VEC; ++k) process(vv.v[k], (int64_t)lv.v[k], 0, i * VEC + k);
    }
  }
  /* tail */
  for (int64_t i = nvec * VEC + gtid; i < n; i += stride) {
    process(values[i], (int64_t)labels[i], twolab ? (int64_t)labels2[i] : 0, i);
  }

  __syncthreads();
  /* flush bins to this block's slab section with plain coalesced stores */
  char* my = slab + (int64_t)blockIdx.x * lay.bytes;
  for (int64_t g = tid; g < ngroups; g += blockDim.x) {
    if (OPS & (B_SUM | B_SSD | B_PROD | B_IDXMIN | B_IDXMAX)) ((SumT*)(my + lay.sum_off))[g] = s_sum[g];
    if (OPS & B_CNT) ((uint32_t*)(my + lay.cnt_off))[g] = s_cnt[g];
    if (OPS & B_PRESENT) ((uint32_t*)(my + lay.present_off))[g] = s_present[g];
    if (OPS & (B_MIN | B_MAX)) ((Enc*)(my + lay.minmax_off))[g] = s_mm[g];
    if (OPS & B_NANFLAG) ((uint32_t*)(my + lay.nanflag_off))[g] = s_nanflag[g];
  }
}

/* ---- kernel 2: combine per-block slabs ----------------------------------
 * grid.y splits the slab-block range so small group counts still fill the
 * chip; with gridDim.y == 1 final bins are written directly (min/max
 * decoded inline), otherwise each chunk folds its share and merges into
 * memset-initialized bins with atomics (min/max stay encoded for k_decode). */
template <typename V, int OPS>
__global__ void k_combine(const char* __restrict__ slab, int nblocks,
                          int64_t ngroups, BinLayout lay, void* out_sum,
                          int64_t* out_count, uint32_t* out_present,
                          void* out_min, void* out_max, uint32_t* out_nanflag) {
  using TR = Traits<V>;
  using Acc = typename TR::Acc;
  using SumT = typename std::conditional<
      (OPS & (B_SSD | B_WELFORD)) != 0, double,
      typename std::conditional<(OPS & (B_IDXMIN | B_IDXMAX)) != 0, int64_t, Acc>::type>::type;
  using Enc = typename TR::Enc;
  constexpr bool IS_PROD = (OPS & B_PROD) != 0;
  const int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= ngroups) return;
  const int S = gridDim.y;
  const int chunk = (nblocks + S - 1) / S;
  const int b0 = blockIdx.y * chunk;
  int b1 = b0 + chunk;
  if (b1 > nblocks) b1 = nblocks;
  if (b0 >= b1) return;

  SumT s = IS_PROD ? (SumT)1 : (SumT)0;
  if (OPS & B_IDXMIN) s = (SumT)INT64_MAX;
  if (OPS & B_IDXMAX) s = (SumT)(-1);
  int64_t c = 0;
  uint32_t p = 0, nf = 0;
  double w_sum = 0.0;
  Enc mn = (Enc)~(Enc)0, mx = (Enc)0;
  for (int b = b0; b < b1; ++b) {
    const char* blk = slab + (int64_t)b * lay.bytes;
    if (OPS & (B_SUM | B_SSD)) s += ((const SumT*)(blk + lay.sum_off))[g];
    if (OPS & B_WELFORD) {
      /* pairwise var merge: fold ssd_i + sum_i^2/n_i, subtract the global
       * term after the loop (the reference's _var_combine adjustment,
       * aggregations.py:392-451, in closed form) */
      const double ssd_i = ((const double*)(blk + lay.sum_off))[g];
      const double sum_i = ((const double*)(blk + lay.sumx_off))[g];
      const double n_i = (double)((const uint32_t*)(blk + lay.cnt_off))[g];
      s += (SumT)(ssd_i + (n_i > 0.0 ? (sum_i * sum_i) / n_i : 0.0));
      w_sum += sum_i;
    }
    if (IS_PROD) s *= ((const SumT*)(blk + lay.sum_off))[g];
    if (OPS & B_IDXMIN) {
      const SumT x = ((const SumT*)(blk + lay.sum_off))[g];
      s = x < s ? x : s;
    }
    if (OPS & B_IDXMAX) {
      const SumT x = ((const SumT*)(blk + lay.sum_off))[g];
      s = x > s ? x : s;
    }
    if (OPS & B_CNT) c += (int64_t)((const uint32_t*)(blk + lay.cnt_off))[g];
    if (OPS & B_PRESENT) p |= ((const uint32_t*)(blk + lay.present_off))[g];
    if (OPS & B_MIN) {
      Enc e = ((const Enc*)(blk + lay.minmax_off))[g];
      mn = e < mn ? e : mn;
    }
    if (OPS & B_MAX) {
      Enc e = ((const Enc*)(blk + lay.minmax_off))[g];
      mx = e > mx ? e : mx;
    }
    if (OPS & B_NANFLAG) nf |= ((const uint32_t*)(blk + lay.nanflag_off))[g];
  }
  if (S == 1) {
    const bool present = (OPS & B_PRESENT) ? (p != 0) : (c != 0);
    if (OPS & B_WELFORD) {
      const double total = (double)s - (c > 0 ? (w_sum * w_sum) / (double)c : 0.0);
      ((double*)out_sum)[g] = c > 0 ? total : 0.0;
      ((double*)out_min)[g] = w_sum;
      out_count[g] = c;
      return;
    }
    if (OPS & (B_SUM | B_SSD | B_PROD | B_IDXMIN | B_IDXMAX)) ((SumT*)out_sum)[g] = s;
    if (OPS & B_CNT) out_count[g] = c;
    if (OPS & B_PRESENT) out_present[g] = p;
    if (OPS & B_MIN) ((V*)out_min)[g] = present ? TR::dec(mn) : TR::pos_inf();
    if (OPS & B_MAX) ((V*)out_max)[g] = present ? TR::dec(mx) : (V)-TR::pos_inf();
    if (OPS & B_NANFLAG) out_nanflag[g] = nf;
  } else {
    if (OPS & (B_SUM | B_SSD)) acc_add((SumT*)out_sum + g, s);
    if (IS_PROD) {
      if (s != (SumT)1) acc_mul((SumT*)out_sum + g, s);
    }
    if (OPS & B_IDXMIN) idx_min((int64_t*)out_sum + g, (int64_t)s);
    if (OPS & B_IDXMAX) idx_max((int64_t*)out_sum + g, (int64_t)s);
    if ((OPS & B_CNT) && c)
      atomicAdd(reinterpret_cast<unsigned long long*>(&out_count[g]), (unsigned long long)c);
    if ((OPS & B_PRESENT) && p) out_present[g] = 1u;
    if (OPS & B_MIN) enc_min(&((Enc*)out_min)[g], mn);
    if (OPS & B_MAX) enc_max(&((Enc*)out_max)[g], mx);
    if ((OPS & B_NANFLAG) && nf) out_nanflag[g] = 1u;
  }
}

/* ---- global-atomic path -------------------------------------------------- */
/* bins live in HBM/LLC; min/max bins are ENCODED in out_min/out_max during
 * accumulation (memset-friendly init) and decoded by k_decode afterwards. */
template <typename V, typename L, int OPS>
__launch_bounds__(BLOCK_ATOMIC) __global__ void k_reduce_atomic(
    const V* __restrict__ values, const L* __restrict__ labels,
    const L* __restrict__ labels2, int64_t n, int64_t ngroups, int64_t g0,
    int64_t g1, const double* __restrict__ means,
    const V* __restrict__ target, int64_t row_offset, int skipnan,
    void* out_sum, int64_t* out_count, uint32_t* out_present, void* out_min,
    void* out_max, uint32_t* out_nanflag) {
  using TR = Traits<V>;
  using Acc = typename TR::Acc;
  using Enc = typename TR::Enc;
  constexpr int VEC = TR::VEC;
  constexpr bool IS_PROD = (OPS & B_PROD) != 0;
  const bool twolab = labels2 != nullptr;

  auto process = [&](V v, int64_t l0raw, int64_t l1raw, int64_t row) {
    uint64_t code;
    if (twolab) {
      if ((uint64_t)l0raw >= (uint64_t)g0 || (uint64_t)l1raw >= (uint64_t)g1) return;
      code = (uint64_t)l0raw * (uint64_t)g1 + (uint64_t)l1raw;
    } else {
      if ((uint64_t)l0raw >= (uint64_t)ngroups) return;
      code = (uint64_t)l0raw;
    }
    const bool vnan = TR::isnan_(v);
    if (OPS & B_PRESENT) out_present[code] = 1u;
    if (vnan && skipnan) return;
    if (OPS & B_SUM) acc_add(&((Acc*)out_sum)[code], (Acc)v);
    if (IS_PROD) acc_mul(&((Acc*)out_sum)[code], (Acc)v);
    if (OPS & B_SSD) {
      double d = (double)v - means[code];
      atomicAdd(&((double*)out_sum)[code], d * d);
    }
    if (OPS & (B_IDXMIN | B_IDXMAX)) {
      bool match = true;
      if (target) {
        const V t = target[code];
        match = vnan ? TR::isnan_(t) : (!TR::isnan_(t) && v == t);
      }
      if (match) {
        if (OPS & B_IDXMIN) idx_min((int64_t*)out_sum + code, row + row_offset);
        if (OPS & B_IDXMAX) idx_max((int64_t*)out_sum + code, row + row_offset);
      }
    }
    if (OPS & B_CNT) {
      if (!vnan)
        atomicAdd(reinterpret_cast<unsigned long long*>(&out_count[code]), 1ull);
    }
    if (OPS & (B_MIN | B_MAX)) {
      if (vnan) {
        if (OPS & B_NANFLAG) out_nanflag[code] = 1u;
      } else {
        if (OPS & B_MIN) enc_min(&((Enc*)out_min)[code], TR::enc(v));
        if (OPS & B_MAX) enc_max(&((Enc*)out_max)[code], TR::enc(v));
      }
    }
  };

  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int64_t gtid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t nvec = n / VEC;
  for (int64_t i = gtid; i < nvec; i += stride) {
    Vec<V, VEC> vv = *reinterpret_cast<const Vec<V, VEC>*>(values + i * VEC);
    Vec<L, VEC> lv = *reinterpret_cast<const Vec<L, VEC>*>(labels + i * VEC);
    if (twolab) {
      Vec<L, VEC> lv2 = *reinterpret_cast<const Vec<L, VEC>*>(labels2 + i * VEC);
#pragma unroll
      for (int k = 0; k < VEC; ++k)
        process(vv.v[k], (int64_t)lv.v[k], (int64_t)lv2.v[k], i * VEC + k);
    } else {
#pragma unroll
      for (int k = 0; k < VEC; ++k) process(vv.v[k], (int64_t)lv.v[k], 0, i * VEC + k);
    }
  }
  for (int64_t i = nvec * VEC + gtid; i < n; i += stride) {
    process(values[i], (int64_t)labels[i], twolab ? (int64_t)labels2[i] : 0, i);
  }
}


/* ---- column path: grouped reduce over a strided/leading axis ------------- */
/* For array (..., reduced axis) with leading dims — e.g. the hour-of-day
 * climatology (reference asv ERA5 shapes, BASELINE config 4). The label
 * vector lives on the (small) reduced axis only, so the host argsorts it
 * once (the GPU analogue of the reference's _prepare_for_flox sort,
 * aggregate_flox.py:9-23 — but over N_t labels, not N_t*M rows) and the
 * kernel walks rows in GROUP ORDER: each thread owns one column, keeps ONE
 * running accumulator in registers, and writes it out at each (wave-uniform)
 * segment boundary. No atomics, no LDS bins, any ngroups.
 * Element (t, c) is at values[t*ldm + c] (column stride 1 — the natural
 * layout for time-major climate data; a torch permute view, no transpose).
 * Outputs are (ngroups, m) group-major so flush stores coalesce. */
constexpr int COLS_BLOCK = 256;
constexpr int COLS_TTILE = 2048;

/* VC columns per thread (16-B vector loads per G13); blockIdx.y = row chunk.
 * With several row chunks each chunk writes PARTIAL bins (cnt as u32) into
 * its section of the scratch slab, identical in layout to the 1-D path's
 * per-block slab, and k_combine folds the chunks; with one chunk the final
 * bins are written directly (cnt as i64). */
template <typename V, int OPS, int VC, bool SLAB, bool SKIP>
__launch_bounds__(COLS_BLOCK) __global__ void k_reduce_cols(
    const V* __restrict__ values, const int* __restrict__ codes_sorted,
    const int* __restrict__ perm, int64_t n_t, int64_t m, int64_t ldm,
    int64_t ngroups, const double* __restrict__ means, int skipnan,
    int64_t chunk_rows, const int64_t* __restrict__ chunk_offs,
    char* __restrict__ slab, BinLayout lay,
    void* out_sum, int64_t* out_count, uint32_t* out_present, void* out_min,
    void* out_max, uint32_t* out_nanflag) {
  using TR = Traits<V>;
  using Acc = typename TR::Acc;
  using SumT = typename std::conditional<
      (OPS & (B_SSD | B_WELFORD)) != 0, double,
      typename std::conditional<(OPS & (B_IDXMIN | B_IDXMAX)) != 0, int64_t, Acc>::type>::type;
  using Enc = typename TR::Enc;
  constexpr bool IS_PROD = (OPS & B_PROD) != 0;

  __shared__ int s_code[COLS_TTILE];
  __shared__ int s_perm[COLS_TTILE];

  const int64_t c0 = ((int64_t)blockIdx.x * COLS_BLOCK + threadIdx.x) * VC;
  const bool full = c0 + VC <= m;
  int64_t t_begin, t_end;
  if (chunk_offs) {
    t_begin = chunk_offs[blockIdx.y];
    t_end = chunk_offs[blockIdx.y + 1];
  } else {
    t_begin = (int64_t)blockIdx.y * chunk_rows;
    t_end = (t_begin + chunk_rows < n_t) ? t_begin + chunk_rows : n_t;
  }

  char* my_slab = SLAB ? slab + (int64_t)blockIdx.y * lay.bytes : nullptr;

  constexpr bool IS_WEL = (OPS & B_WELFORD) != 0;
  SumT acc[VC];
  uint32_t cnt[VC];
  Enc mn[VC], mx[VC];
  uint32_t nanflag[VC];
  double mean_g[VC];
  /* shifted-variance state: s1 = sum(x - x0), acc holds s2 = sum((x-x0)^2) */
  double w_s1[VC], w_x0[VC];
  uint32_t w_have[VC];
  int cur_g = -1;

  auto reset = [&]() {
#pragma unroll
    for (int k = 0; k < VC; ++k) {
      acc[k] = IS_PROD ? (SumT)1 : (SumT)0;
      cnt[k] = 0;
      mn[k] = (Enc)~(Enc)0;
      mx[k] = (Enc)0;
      nanflag[k] = 0;
      if (IS_WEL) {
        w_s1[k] = 0.0;
        w_x0[k] = 0.0;
        w_have[k] = 0;
      }
    }
  };
  reset();

  auto flush = [&](int g) {
    if (g < 0) return;
#pragma unroll
    for (int k = 0; k < VC; ++k) {
      if (c0 + k >= m) break;
      const int64_t o = (int64_t)g * m + c0 + k;
      if (IS_WEL) {
        /* convert shifted sums to the var_chunk triple: sum = s1 + n*x0,
         * ssd = s2 - s1^2/n (the mean-shifted form) */
        const double nn = (double)cnt[k];
        const double sum = w_s1[k] + nn * w_x0[k];
        const double ssd = cnt[k] ? (double)acc[k] - (w_s1[k] * w_s1[k]) / nn : 0.0;
        if (SLAB) {
          ((double*)(my_slab + lay.sum_off))[o] = ssd;
          ((double*)(my_slab + lay.sumx_off))[o] = sum;
          ((uint32_t*)(my_slab + lay.cnt_off))[o] = cnt[k];
        } else {
          ((double*)out_sum)[o] = ssd;
          ((double*)out_min)[o] = sum; /* sum-of-x rides the out_min slot */
          out_count[o] = (int64_t)cnt[k];
        }
        continue;
      }
      if (SLAB) {
        if (OPS & (B_SUM | B_SSD | B_PROD)) ((SumT*)(my_slab + lay.sum_off))[o] = acc[k];
        if (OPS & B_CNT) ((uint32_t*)(my_slab + lay.cnt_off))[o] = cnt[k];
        if (OPS & B_PRESENT) ((uint32_t*)(my_slab + lay.present_off))[o] = 1u;
        if (OPS & B_MIN) ((Enc*)(my_slab + lay.minmax_off))[o] = mn[k];
        if (OPS & B_MAX) ((Enc*)(my_slab + lay.minmax_off))[o] = mx[k];
        if (OPS & B_NANFLAG) ((uint32_t*)(my_slab + lay.nanflag_off))[o] = nanflag[k];
      } else {
        if (OPS & (B_SUM | B_SSD | B_PROD)) ((SumT*)out_sum)[o] = acc[k];
        if (OPS & B_CNT) out_count[o] = (int64_t)cnt[k];
        if (OPS & B_PRESENT) out_present[o] = 1u;
        if (OPS & B_MIN) ((Enc*)out_min)[o] = mn[k];
        if (OPS & B_MAX) ((Enc*)out_max)[o] = mx[k];
        if (OPS & B_NANFLAG) out_nanflag[o] = nanflag[k];
      }
    }
  };

  auto consume = [&](V v, int k, bool lanes_ok) {
    if (!lanes_ok) return;
    /* SKIP is compile-time (skipnan-templated variants): the skip branch
     * and the count select drop out of the per-element stream */
    const bool vnan = TR::isnan_(v);
    if constexpr (SKIP) {
      if (vnan) return;
    }
    if (IS_WEL) {
      if (!w_have[k]) {
        w_x0[k] = (double)v; /* non-skip: a NaN first value poisons the group */
        w_have[k] = 1;
      }
      const double d = (double)v - w_x0[k];
      w_s1[k] += d;
      acc[k] += (SumT)(d * d);
      if constexpr (SKIP) cnt[k] += 1u; else cnt[k] += vnan ? 0u : 1u;
      return;
    }
    if (OPS & B_SUM) acc[k] += (SumT)v;
    if (IS_PROD) acc[k] *= (SumT)v;
    if (OPS & B_SSD) {
      const double d = (double)v - mean_g[k];
      acc[k] += d * d;
    }
    if (OPS & B_CNT) {
      if constexpr (SKIP) cnt[k] += 1u; else cnt[k] += vnan ? 0u : 1u;
    }
    if (OPS & (B_MIN | B_MAX)) {
      if (vnan) {
        if (OPS & B_NANFLAG) nanflag[k] = 1u;
      } else {
        const Enc e = TR::enc(v);
        if (OPS & B_MIN) mn[k] = e < mn[k] ? e : mn[k];
        if (OPS & B_MAX) mx[k] = e > mx[k] ? e : mx[k];
      }
    }
  };

  for (int64_t t0 = t_begin; t0 < t_end; t0 += COLS_TTILE) {
    const int nt = (int)((t_end - t0 < COLS_TTILE) ? (t_end - t0) : COLS_TTILE);
    __syncthreads();
    for (int i = threadIdx.x; i < nt; i += COLS_BLOCK) {
      s_code[i] = codes_sorted[t0 + i];
      s_perm[i] = perm[t0 + i];
    }
    __syncthreads();
    const bool col_act = c0 < m;
    int i = 0;
    while (i < nt) {
      const int g = s_code[i];
      if (g != cur_g) {
        flush(cur_g);
        reset();
        cur_g = g;
        if ((OPS & B_SSD) && g >= 0 && col_act) {
#pragma unroll
          for (int k = 0; k < VC; ++k)
            if (c0 + k < m) mean_g[k] = means[(int64_t)g * m + c0 + k];
        }
      }
      if (g < 0 || !col_act) {
        ++i;
        continue;
      }
      /* two rows of the same segment per iteration: halves the LDS reads,
       * compares and loop control per element (segment boundaries are rare
       * relative to rows) */
      if (VC > 1 && full) {
        if (i + 1 < nt && s_code[i + 1] == g) {
          const int64_t rowa = (int64_t)s_perm[i] * ldm;
          const int64_t rowb = (int64_t)s_perm[i + 1] * ldm;
          Vec<V, VC> va = *reinterpret_cast<const Vec<V, VC>*>(values + rowa + c0);
          Vec<V, VC> vb = *reinterpret_cast<const Vec<V, VC>*>(values + rowb + c0);
#pragma unroll
          for (int k = 0; k < VC; ++k) consume(va.v[k], k, true);
#pragma unroll
          for (int k = 0; k < VC; ++k) consume(vb.v[k], k, true);
          i += 2;
          continue;
        }
        Vec<V, VC> vv = *reinterpret_cast<const Vec<V, VC>*>(values + (int64_t)s_perm[i] * ldm + c0);
#pragma unroll
        for (int k = 0; k < VC; ++k) consume(vv.v[k], k, true);
        ++i;
      } else {
        const int64_t row = (int64_t)s_perm[i] * ldm;
#pragma unroll
        for (int k = 0; k < VC; ++k)
          consume((c0 + k < m) ? values[row + c0 + k] : (V)0, k, c0 + k < m);
        ++i;
      }
    }
  }
  flush(cur_g);
}

__global__ void k_init_cursors(uint32_t* cur, int n, uint32_t cap) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) cur[i] = (uint32_t)i * cap;
}

/* init product bins to 1 (memset cannot) */
__global__ void k_fill_f64(double* p, int64_t n, double v) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = v;
}
__global__ void k_fill_i64(int64_t* p, int64_t n, int64_t v) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = v;
}

/* decode min/max bins in place (global-atomic path) */
template <typename V, int OPS>
__global__ void k_decode(int64_t ngroups, void* out_min, void* out_max,
                         const int64_t* out_count, const uint32_t* out_present) {
  using TR = Traits<V>;
  using Enc = typename TR::Enc;
  const int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= ngroups) return;
  const bool present =
      (OPS & B_PRESENT) ? (out_present[g] != 0) : (out_count && out_count[g] != 0);
  if (OPS & B_MIN) {
    Enc e = ((Enc*)out_min)[g];
    ((V*)out_min)[g] = present ? TR::dec(e) : TR::pos_inf();
  }
  if (OPS & B_MAX) {
    Enc e = ((Enc*)out_max)[g];
    ((V*)out_max)[g] = present ? TR::dec(e) : (V)-TR::pos_inf();
  }
}


/* ---- partition path (huge ngroups, e.g. 1e7): bucket scatter + per-bucket
 * LDS reduce — the GPU analogue of the reference's sort-by-key + reduceat
 * engine (aggregate_flox.py:9-23 _prepare_for_flox + :133-192
 * _np_grouped_op), except only ONE radix digit is needed: rows are
 * partitioned by the high bits of their group code into buckets whose bins
 * fit in LDS, then each bucket is reduced like the LDS path. Traffic:
 * labels (count) + values+labels (scatter read) + pairs (write+read) ~=
 * 3.7x the algorithmic bytes, vs ~45x slower raw global atomics. */

template <typename V> struct PairT;
template <> struct PairT<float> { float v; uint32_t lc; };
template <> struct PairT<int32_t> { int32_t v; uint32_t lc; };
template <> struct PairT<double> { double v; uint32_t lc; uint32_t pad; };
template <> struct PairT<int64_t> { int64_t v; uint32_t lc; uint32_t pad; };

__device__ __forceinline__ int64_t code_of(int64_t l0, int64_t l1, bool twolab,
                                           int64_t g0, int64_t g1, int64_t ngroups) {
  if (twolab) {
    if ((uint64_t)l0 >= (uint64_t)g0 || (uint64_t)l1 >= (uint64_t)g1) return -1;
    return l0 * g1 + l1;
  }
  return ((uint64_t)l0 >= (uint64_t)ngroups) ? -1 : l0;
}

template <typename L>
__launch_bounds__(256) __global__ void k_part_count(
    const L* __restrict__ labels, const L* __restrict__ labels2, int64_t n,
    int64_t ngroups, int64_t g0, int64_t g1, int shift, int B,
    uint32_t* __restrict__ bucket_counts) {
  extern __shared__ __attribute__((aligned(16))) char smem_pc[];
  uint32_t* s_hist = (uint32_t*)smem_pc;
  for (int i = threadIdx.x; i < B; i += blockDim.x) s_hist[i] = 0;
  __syncthreads();
  const bool twolab = labels2 != nullptr;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    int64_t code = code_of((int64_t)labels[i], twolab ? (int64_t)labels2[i] : 0,
                           twolab, g0, g1, ngroups);
    if (code >= 0) atomicAdd(&s_hist[(int)(code >> shift)], 1u);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < B; i += blockDim.x)
    if (s_hist[i]) atomicAdd(&bucket_counts[i], s_hist[i]);
}

/* tile-staged scatter: per tile, histogram -> tiny scan -> global
 * reservation (one returning atomic per nonempty bucket) -> bucket-ordered
 * staging in LDS with per-slot destinations -> coalesced dump.
 * With the two-level scheme this pass NEVER sees more than PART_SUB=64
 * buckets (pass A uses the <= 64 super-buckets; single-level runs only when
 * the fine bucket count is <= 64), so the histogram/scan state is tiny and
 * three blocks fit per CU. */
constexpr int PART_BLOCK = 512;

/* per-wave-histogram tile scatter machinery, shared by both scatter passes:
 * each of the NW waves owns its own 64-entry histogram slice (layout
 * [bucket][wave]) so LDS atomic contention is divided by NW, the exclusive
 * scan over the 512 (bucket, wave) slots runs as wave shuffle scans plus
 * one tiny cross-wave combine (2 barriers instead of 18), and the scanned
 * slots double as the staging cursors. */
constexpr int PART_NW = PART_BLOCK / 64;

__device__ __forceinline__ uint32_t wave_incl_scan(uint32_t v) {
#pragma unroll
  for (int d = 1; d < 64; d <<= 1) {
    const uint32_t o = __shfl_up((int)v, d);
    if ((threadIdx.x & 63) >= d) v += o;
  }
  return v;
}

template <typename V, typename L, bool CROW = false,
          int T = (sizeof(V) == 4 ? 6144 : 3072)>
__launch_bounds__(PART_BLOCK) __global__ void k_part_scatter(
    const V* __restrict__ values, const L* __restrict__ labels,
    const L* __restrict__ labels2, int64_t n, int64_t ngroups, int64_t g0,
    int64_t g1, int shift, int B, int Bpad /* unused, <= 64 buckets */,
    uint32_t* __restrict__ cursors, uint32_t cap /* 0 = exact bases */,
    uint32_t* __restrict__ overflow, PairT<V>* __restrict__ pairs,
    int64_t row_base = 0 /* CROW: global row = row_base + i in .pad */) {
  constexpr int RPT = T / PART_BLOCK;
  constexpr int NB = 64;
  constexpr int NE = NB * PART_NW; /* = PART_BLOCK slots */
  extern __shared__ __attribute__((aligned(16))) char smem_ps[];
  uint32_t* s_slot = (uint32_t*)smem_ps;              /* [NE] hist -> excl -> cursor */
  uint32_t* s_wtot = s_slot + NE;                     /* [PART_NW + 1] wave totals */
  uint32_t* s_gbase = s_wtot + PART_NW + 1;           /* [NB] gbase - bucket excl */
  uint32_t* s_dest = s_gbase + NB + 3;                /* [T]; pads stage to 16 B */
  PairT<V>* s_stage = (PairT<V>*)(s_dest + T);        /* [T] */

  const bool twolab = labels2 != nullptr;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;

  for (int64_t tile = (int64_t)blockIdx.x * T; tile < n; tile += (int64_t)gridDim.x * T) {
    const int nt = (int)((n - tile < T) ? (n - tile) : T);
    s_slot[tid] = 0;
    __syncthreads();

    V rv[RPT];
    uint32_t rlc[RPT];
    int rbk[RPT];
    /* per-thread CONTIGUOUS rows -> 16-B vector loads (G13) */
    const int base = tid * RPT;
    if (base + RPT <= nt) {
      constexpr int VW = 16 / (int)sizeof(V);
#pragma unroll
      for (int k = 0; k < RPT; k += VW) {
        const Vec<V, VW> vv =
            *reinterpret_cast<const Vec<V, VW>*>(values + tile + base + k);
#pragma unroll
        for (int j = 0; j < VW; ++j) rv[k + j] = vv.v[j];
      }
      Vec<L, 2> lv[RPT / 2 > 0 ? RPT / 2 : 1];
      if (sizeof(L) == 8) {
#pragma unroll
        for (int k = 0; k < RPT; k += 2)
          lv[k / 2] = *reinterpret_cast<const Vec<L, 2>*>(labels + tile + base + k);
      }
#pragma unroll
      for (int k = 0; k < RPT; ++k) {
        const int64_t i = tile + base + k;
        const int64_t l0 = sizeof(L) == 8 ? (int64_t)lv[k / 2].v[k & 1] : (int64_t)labels[i];
        const int64_t code = code_of(l0, twolab ? (int64_t)labels2[i] : 0, twolab,
                                     g0, g1, ngroups);
        rbk[k] = -1;
        if (code >= 0) {
          rbk[k] = (int)(code >> shift);
          rlc[k] = (uint32_t)(code - ((int64_t)rbk[k] << shift));
          atomicAdd(&s_slot[wid * NB + rbk[k]], 1u);
        }
      }
    } else {
#pragma unroll
      for (int k = 0; k < RPT; ++k) {
        const int idx = base + k;
        rbk[k] = -1;
        if (idx < nt) {
          const int64_t i = tile + idx;
          const int64_t code = code_of((int64_t)labels[i],
                                       twolab ? (int64_t)labels2[i] : 0, twolab,
                                       g0, g1, ngroups);
          if (code >= 0) {
            rv[k] = values[i];
            rbk[k] = (int)(code >> shift);
            rlc[k] = (uint32_t)(code - ((int64_t)rbk[k] << shift));
            atomicAdd(&s_slot[wid * NB + rbk[k]], 1u);
          }
        }
      }
    }
    __syncthreads();
    /* exclusive scan over the NE slots: wave shuffle scans + wave totals */
    {
      /* s_slot layout is [wave][bucket] (addr = w*NB + b): the per-row
       * histogram/cursor atomics touch all 32 LDS banks (the previous
       * [bucket][wave] stride-NW layout mapped random buckets onto only 4
       * banks). Measured NEUTRAL on the 1e9/1e7 sum (10.60 vs 10.62 ms) —
       * the wave parking is memory-latency-bound, not LDS-bound — kept for
       * the clean banking. The scan is still over the semantic
       * (bucket, wave) order = tid; only this once-per-tile gather reads
       * transposed. */
      const int tb = tid / PART_NW, tw = tid % PART_NW;
      const uint32_t mine = s_slot[tw * NB + tb];
      const uint32_t incl = wave_incl_scan(mine);
      if ((tid & 63) == 63) s_wtot[wid] = incl;
      __syncthreads();
      if (tid == 0) {
        uint32_t run = 0;
        for (int w = 0; w < PART_NW; ++w) {
          const uint32_t x = s_wtot[w];
          s_wtot[w] = run;
          run += x;
        }
        s_wtot[PART_NW] = run;
      }
      __syncthreads();
      s_slot[tw * NB + tb] = incl - mine + s_wtot[wid];
    }
    __syncthreads();
    const uint32_t total = s_wtot[PART_NW];
    for (int b = tid; b < B; b += PART_BLOCK) {
      const uint32_t excl = s_slot[b]; /* (b, w=0) in [w][b] layout */
      const uint32_t nxt = (b + 1 < NB) ? s_slot[b + 1] : total;
      const uint32_t cnt = nxt - excl;
      if (cnt) {
        uint32_t gb = atomicAdd(&cursors[b], cnt);
        if (cap && gb + cnt > (uint32_t)(b + 1) * cap) {
          /* optimistic region overflow: flag it and keep writes in-bounds
           * (results are discarded and recomputed by the exact path) */
          *overflow = 1u;
          gb = (uint32_t)b * cap;
        }
        s_gbase[b] = gb - excl;
      }
    }
    __syncthreads();
#pragma unroll
    for (int k = 0; k < RPT; ++k) {
      if (rbk[k] >= 0) {
        const uint32_t pos = atomicAdd(&s_slot[wid * NB + rbk[k]], 1u);
        s_stage[pos].v = rv[k];
        s_stage[pos].lc = rlc[k];
        if constexpr (CROW && sizeof(PairT<V>) == 16)
          s_stage[pos].pad = (uint32_t)(tile + base + k + row_base);
        s_dest[pos] = s_gbase[rbk[k]] + pos;
      }
    }
    __syncthreads();
    for (int i = tid; i < (int)total; i += PART_BLOCK) pairs[s_dest[i]] = s_stage[i];
    __syncthreads();
  }
}

/* second-level scatter: rows of one super-bucket (already contiguous in
 * `in`) are partitioned into their <= 64 fine buckets. Same tile machinery
 * as k_part_scatter, with a fixed 64-entry histogram. */
constexpr int PART_SUB = 64;

template <typename V, bool CROW = false,
          int T = (sizeof(V) == 4 ? 6144 : 3072)>
__launch_bounds__(PART_BLOCK) __global__ void k_part_scatter2(
    const PairT<V>* __restrict__ in, const uint32_t* __restrict__ baseA,
    uint32_t capA /* 0: baseA[sb]..baseA[sb+1]; else sb*capA..baseA[sb] */,
    int shift /* fine-bucket shift */, int bfine /* total fine buckets */,
    uint32_t* __restrict__ cursors,
    uint32_t cap2 /* 0 = exact fine bases */, uint32_t* __restrict__ overflow,
    PairT<V>* __restrict__ out) {
  constexpr int RPT = T / PART_BLOCK;
  constexpr int NB = PART_SUB;
  constexpr int NE = NB * PART_NW;
  extern __shared__ __attribute__((aligned(16))) char smem_p2[];
  uint32_t* s_slot = (uint32_t*)smem_p2;
  uint32_t* s_wtot = s_slot + NE;
  uint32_t* s_gbase = s_wtot + PART_NW + 1;
  uint32_t* s_dest = s_gbase + NB + 3;
  PairT<V>* s_stage = (PairT<V>*)(s_dest + T);

  const int sb = blockIdx.y;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int64_t r0 = capA ? (int64_t)(uint32_t)sb * capA : (int64_t)baseA[sb];
  int64_t r1 = capA ? (int64_t)baseA[sb] : (int64_t)baseA[sb + 1];
  /* pass-A overflow leaves its cursor beyond the capacity region; clamp so
   * we never read the neighbouring region's (or uninitialized) pairs — the
   * results are discarded and recomputed by the exact path anyway */
  if (capA && r1 > r0 + (int64_t)capA) r1 = r0 + (int64_t)capA;
  const uint32_t lmask = (1u << shift) - 1u;

  for (int64_t tile = r0 + (int64_t)blockIdx.x * T; tile < r1;
       tile += (int64_t)gridDim.x * T) {
    const int nt = (int)((r1 - tile < T) ? (r1 - tile) : T);
    s_slot[tid] = 0;
    __syncthreads();
    V rv[RPT];
    uint32_t rlc[RPT];
    uint32_t rpd[CROW ? RPT : 1];
    int rbk[RPT];
    const int base = tid * RPT;
#pragma unroll
    for (int k = 0; k < RPT; ++k) {
      const int idx = base + k;
      rbk[k] = -1;
      if (idx < nt) {
        const PairT<V> pr = in[tile + idx];
        rv[k] = pr.v;
        if constexpr (CROW && sizeof(PairT<V>) == 16) rpd[k] = pr.pad;
        /* mask: stale pairs after a pass-A overflow can carry garbage lc */
        rbk[k] = (int)((pr.lc >> shift) & (PART_SUB - 1));
        rlc[k] = pr.lc & lmask;
        atomicAdd(&s_slot[wid * NB + rbk[k]], 1u);
      }
    }
    __syncthreads();
    {
      /* s_slot layout is [wave][bucket] (addr = w*NB + b): the per-row
       * histogram/cursor atomics touch all 32 LDS banks (the previous
       * [bucket][wave] stride-NW layout mapped random buckets onto only 4
       * banks). Measured NEUTRAL on the 1e9/1e7 sum (10.60 vs 10.62 ms) —
       * the wave parking is memory-latency-bound, not LDS-bound — kept for
       * the clean banking. The scan is still over the semantic
       * (bucket, wave) order = tid; only this once-per-tile gather reads
       * transposed. */
      const int tb = tid / PART_NW, tw = tid % PART_NW;
      const uint32_t mine = s_slot[tw * NB + tb];
      const uint32_t incl = wave_incl_scan(mine);
      if ((tid & 63) == 63) s_wtot[wid] = incl;
      __syncthreads();
      if (tid == 0) {
        uint32_t run = 0;
        for (int w = 0; w < PART_NW; ++w) {
          const uint32_t x = s_wtot[w];
          s_wtot[w] = run;
          run += x;
        }
        s_wtot[PART_NW] = run;
      }
      __syncthreads();
      s_slot[tw * NB + tb] = incl - mine + s_wtot[wid];
    }
    __syncthreads();
    const uint32_t total = s_wtot[PART_NW];
    for (int b = tid; b < NB; b += PART_BLOCK) {
      const uint32_t excl = s_slot[b]; /* (b, w=0) in [w][b] layout */
      const uint32_t nxt = (b + 1 < NB) ? s_slot[b + 1] : total;
      const uint32_t cnt = nxt - excl;
      if (cnt) {
        const uint32_t fb = (uint32_t)sb * PART_SUB + (uint32_t)b;
        if (fb >= (uint32_t)bfine) {
          /* only reachable via pass-A overflow garbage: flag, park the
           * rows at the buffer start (results are discarded) */
          *overflow = 1u;
          s_gbase[b] = 0u - excl;
        } else {
          uint32_t gb = atomicAdd(&cursors[fb], cnt);
          if (cap2 && gb + cnt > (fb + 1u) * cap2) {
            *overflow = 1u;
            gb = fb * cap2;
          }
          s_gbase[b] = gb - excl;
        }
      }
    }
    __syncthreads();
#pragma unroll
    for (int k = 0; k < RPT; ++k) {
      if (rbk[k] >= 0) {
        const uint32_t pos = atomicAdd(&s_slot[wid * NB + rbk[k]], 1u);
        s_stage[pos].v = rv[k];
        s_stage[pos].lc = rlc[k];
        if constexpr (CROW && sizeof(PairT<V>) == 16) s_stage[pos].pad = rpd[k];
        s_dest[pos] = s_gbase[rbk[k]] + pos;
      }
    }
    __syncthreads();
    for (int i = tid; i < (int)total; i += PART_BLOCK) out[s_dest[i]] = s_stage[i];
    __syncthreads();
  }
}

/* one-level DIRECT scatter: every fine bucket (up to 4096) is a pass-A
 * target, so the second scatter pass disappears entirely — streamed bytes
 * drop from ~44 B/row (12r+8w, 8r+8w, 8r) to ~28 B/row (12r+8w, 8r).
 * There is no LDS staging: a register-blocked tile (RPT rows per thread)
 * is histogrammed in LDS, each nonempty bucket reserves its span with ONE
 * returning global atomic, and rows store their pair directly to the
 * reserved slot. Stores are short per-bucket runs (~T/B rows ≈ 100 B at
 * 1e7 groups): partial 128-B lines merge in the XCD L2 first (one
 * workgroup's run lands in one L2) and the remainder merges in the
 * 256 MiB memory-side Infinity Cache, whose capacity easily holds every
 * bucket's write frontier (B × ~2 lines ≈ 1 MB) — so the true HBM write
 * traffic stays ≈8 B/row (verify via WRITE_SIZE ratios).
 * Without the 73 KB staging, LDS is just the 16 KB histogram and the
 * kernel runs 3 workgroups/CU instead of 2 — more latency cover for the
 * scatter's pointer-chasing stores. */
constexpr int PBD = 512;

template <typename V, typename L, bool CROW = false>
__launch_bounds__(PBD, 6) __global__ void k_part_scatter_direct(
    const V* __restrict__ values, const L* __restrict__ labels,
    const L* __restrict__ labels2, int64_t n, int64_t ngroups, int64_t g0,
    int64_t g1, int shift, int B, uint32_t* __restrict__ cursors,
    uint32_t cap /* 0 = exact bases preloaded in cursors */,
    uint32_t* __restrict__ overflow, PairT<V>* __restrict__ pairs,
    int64_t row_base = 0) {
  constexpr int RPT = sizeof(V) == 4 ? 24 : 12;
  constexpr int T = PBD * RPT; /* 12288 rows (4-B V) / 6144 (8-B V) */
  extern __shared__ __attribute__((aligned(16))) char smem_pd[];
  uint32_t* s_hist = (uint32_t*)smem_pd; /* [B] counts -> running cursors */
  const bool twolab = labels2 != nullptr;
  const int tid = threadIdx.x;
  const uint32_t lmask = (1u << shift) - 1u;
  constexpr uint32_t INV = 0xFFFFFFFFu;

  for (int64_t tile = (int64_t)blockIdx.x * T; tile < n;
       tile += (int64_t)gridDim.x * T) {
    const int nt = (int)((n - tile < T) ? (n - tile) : T);
    for (int b = tid; b < B; b += PBD) s_hist[b] = 0u;
    __syncthreads();

    V rv[RPT];
    uint32_t rcode[RPT]; /* full code (bucket<<shift | lc), < 2^25; INV = drop */
    const int base = tid * RPT;
    if (base + RPT <= nt) {
      /* per-thread CONTIGUOUS rows -> 16-B vector loads */
      constexpr int VW = 16 / (int)sizeof(V);
#pragma unroll
      for (int k = 0; k < RPT; k += VW) {
        const Vec<V, VW> vv =
            *reinterpret_cast<const Vec<V, VW>*>(values + tile + base + k);
#pragma unroll
        for (int j = 0; j < VW; ++j) rv[k + j] = vv.v[j];
      }
      if (sizeof(L) == 8) {
#pragma unroll
        for (int k = 0; k < RPT; k += 2) {
          const Vec<L, 2> lv =
              *reinterpret_cast<const Vec<L, 2>*>(labels + tile + base + k);
          const int64_t i = tile + base + k;
          const int64_t c0 = code_of((int64_t)lv.v[0],
                                     twolab ? (int64_t)labels2[i] : 0, twolab,
                                     g0, g1, ngroups);
          const int64_t c1 = code_of((int64_t)lv.v[1],
                                     twolab ? (int64_t)labels2[i + 1] : 0,
                                     twolab, g0, g1, ngroups);
          rcode[k] = c0 >= 0 ? (uint32_t)c0 : INV;
          rcode[k + 1] = c1 >= 0 ? (uint32_t)c1 : INV;
          if (c0 >= 0) atomicAdd(&s_hist[rcode[k] >> shift], 1u);
          if (c1 >= 0) atomicAdd(&s_hist[rcode[k + 1] >> shift], 1u);
        }
      } else {
#pragma unroll
        for (int k = 0; k < RPT; ++k) {
          const int64_t i = tile + base + k;
          const int64_t c0 = code_of((int64_t)labels[i],
                                     twolab ? (int64_t)labels2[i] : 0, twolab,
                                     g0, g1, ngroups);
          rcode[k] = c0 >= 0 ? (uint32_t)c0 : INV;
          if (c0 >= 0) atomicAdd(&s_hist[rcode[k] >> shift], 1u);
        }
      }
    } else {
#pragma unroll
      for (int k = 0; k < RPT; ++k) {
        const int idx = base + k;
        rcode[k] = INV;
        if (idx < nt) {
          const int64_t i = tile + idx;
          const int64_t c0 = code_of((int64_t)labels[i],
                                     twolab ? (int64_t)labels2[i] : 0, twolab,
                                     g0, g1, ngroups);
          if (c0 >= 0) {
            rv[k] = values[i];
            rcode[k] = (uint32_t)c0;
            atomicAdd(&s_hist[rcode[k] >> shift], 1u);
          }
        }
      }
    }
    __syncthreads();
    /* one returning global atomic per nonempty bucket reserves the tile's
     * span; s_hist[b] becomes the running intra-tile cursor */
    for (int b = tid; b < B; b += PBD) {
      const uint32_t cnt = s_hist[b];
      if (cnt) {
        uint32_t gb = atomicAdd(&cursors[b], cnt);
        if (cap && gb + cnt > (uint32_t)(b + 1) * cap) {
          /* optimistic region overflow: flag it and keep writes in-bounds
           * (results are discarded and recomputed by the exact path) */
          *overflow = 1u;
          gb = (uint32_t)b * cap;
        }
        s_hist[b] = gb;
      }
    }
    __syncthreads();
#pragma unroll
    for (int k = 0; k < RPT; ++k) {
      if (rcode[k] != INV) {
        const uint32_t pos = atomicAdd(&s_hist[rcode[k] >> shift], 1u);
        PairT<V> pr{};
        pr.v = rv[k];
        pr.lc = rcode[k] & lmask;
        if constexpr (CROW && sizeof(PairT<V>) == 16)
          pr.pad = (uint32_t)(tile + base + k + row_base);
        pairs[pos] = pr;
      }
    }
    __syncthreads();
  }
}

/* sorted-labels direct path: bucket b's rows are the contiguous range
 * [base[b], base[b+1]) of the ORIGINAL arrays (no scatter passes at all —
 * 12 B/row instead of the partition's ~44 B/row). base comes from
 * k_bucket_bounds; the FH_SORTED_LABELS contract guarantees nondecreasing
 * in-range labels. */
template <typename L>
__global__ void k_bucket_bounds(const L* __restrict__ labels, int64_t n,
                                int nb, int shift,
                                uint32_t* __restrict__ base) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b > nb) return;
  if (b == nb) {
    base[nb] = (uint32_t)n;
    return;
  }
  const int64_t target = (int64_t)b << shift;
  int64_t lo = 0, hi = n; /* lower_bound */
  while (lo < hi) {
    const int64_t mid = (lo + hi) >> 1;
    if ((int64_t)labels[mid] < target) lo = mid + 1; else hi = mid;
  }
  base[b] = (uint32_t)lo;
}

template <typename V, typename L, int OPS>
__launch_bounds__(BLOCK_LDS) __global__ void k_reduce_bucket_direct(
    const V* __restrict__ values, const L* __restrict__ labels,
    const uint32_t* __restrict__ base, int64_t chunk, int gpb, int shift,
    int64_t ngroups, const double* __restrict__ means, int skipnan,
    BinLayout lay, void* out_sum, int64_t* out_count, uint32_t* out_present,
    void* out_min, void* out_max, uint32_t* out_nanflag) {
  using TR = Traits<V>;
  using Acc = typename TR::Acc;
  using SumT = typename std::conditional<
      (OPS & (B_SSD | B_WELFORD)) != 0, double,
      typename std::conditional<(OPS & (B_IDXMIN | B_IDXMAX)) != 0, int64_t, Acc>::type>::type;
  using Enc = typename TR::Enc;
  constexpr bool IS_PROD = (OPS & B_PROD) != 0;

  const int b = blockIdx.y;
  const int64_t bkt_begin = base[b];
  const int64_t bkt_end = base[b + 1];
  if (bkt_begin >= bkt_end) return;
  const int64_t gbase = (int64_t)b << shift;
  const int ng_here = (int)(((gbase + gpb) <= ngroups) ? gpb : (ngroups - gbase));

  extern __shared__ __attribute__((aligned(16))) char smem_rd[];
  SumT* s_sum = (OPS & (B_SUM | B_SSD | B_PROD | B_IDXMIN | B_IDXMAX))
                    ? (SumT*)(smem_rd + lay.sum_off) : nullptr;
  uint32_t* s_cnt = (OPS & B_CNT) ? (uint32_t*)(smem_rd + lay.cnt_off) : nullptr;
  uint32_t* s_present = (OPS & B_PRESENT) ? (uint32_t*)(smem_rd + lay.present_off) : nullptr;
  Enc* s_mm = (OPS & (B_MIN | B_MAX)) ? (Enc*)(smem_rd + lay.minmax_off) : nullptr;
  uint32_t* s_nanflag = (OPS & B_NANFLAG) ? (uint32_t*)(smem_rd + lay.nanflag_off) : nullptr;

  const int tid = threadIdx.x;
  for (int g = tid; g < ng_here; g += blockDim.x) {
    if (OPS & (B_SUM | B_SSD)) s_sum[g] = (SumT)0;
    if (IS_PROD) s_sum[g] = (SumT)1;
    if (OPS & B_CNT) s_cnt[g] = 0u;
    if (OPS & B_PRESENT) s_present[g] = 0u;
    if (OPS & B_MIN) s_mm[g] = (Enc)~(Enc)0;
    if (OPS & B_MAX) s_mm[g] = (Enc)0;
    if (OPS & B_NANFLAG) s_nanflag[g] = 0u;
  }
  __syncthreads();

  /* grid-stride over the bucket's row chunks (handles any skew). Sorted
   * labels put a wave's 64 lanes in at most a couple of groups, so naive
   * per-lane LDS atomics serialize 64-way; instead each wave combines its
   * run segments with a shuffle scan and only the last lane of each run
   * touches LDS (one atomic per run per wave). */
  const int lane = tid & 63;
  constexpr uint32_t LC_INVALID = 0xFFFFFFFFu;
  for (int64_t start = bkt_begin + (int64_t)blockIdx.x * chunk; start < bkt_end;
       start += (int64_t)gridDim.x * chunk) {
    const int64_t end = (start + chunk < bkt_end) ? start + chunk : bkt_end;
    for (int64_t i0 = start + (int64_t)(tid & ~63); i0 < end; i0 += blockDim.x) {
      const int64_t i = i0 + lane;
      uint32_t lc = LC_INVALID;
      V v = (V)0;
      bool vnan = false;
      if (i < end) {
        lc = (uint32_t)((int64_t)labels[i] - gbase);
        if (lc >= (uint32_t)ng_here) {
          lc = LC_INVALID; /* defensive: contract says in-range */
        } else {
          v = values[i];
          vnan = TR::isnan_(v);
        }
      }
      const bool skiprow = vnan && skipnan;
      SumT s = (SumT)0;
      uint32_t cn = 0;
      Enc mm = (OPS & B_MIN) ? (Enc)~(Enc)0 : (Enc)0; /* combine identity */
      uint32_t nf = 0;
      if (lc != LC_INVALID) {
        if (OPS & B_SUM) s = skiprow ? (SumT)0 : (SumT)(Acc)v;
        if (IS_PROD) s = skiprow ? (SumT)1 : (SumT)(Acc)v;
        if (OPS & B_SSD) {
          if (!skiprow) {
            const double d = (double)v - means[gbase + lc];
            s = (SumT)(d * d);
          }
        }
        cn = vnan ? 0u : 1u;
        if ((OPS & (B_MIN | B_MAX)) && !vnan) mm = TR::enc(v);
        nf = vnan ? 1u : 0u;
      } else if (IS_PROD) {
        s = (SumT)1;
      }
#pragma unroll
      for (int d = 1; d < 64; d <<= 1) {
        const uint32_t olc = __shfl_up(lc, d, 64);
        const SumT os = __shfl_up(s, d, 64);
        const uint32_t ocn = __shfl_up(cn, d, 64);
        const Enc omm = __shfl_up(mm, d, 64);
        const uint32_t onf = __shfl_up(nf, d, 64);
        if (lane >= d && lc != LC_INVALID && olc == lc) {
          if (OPS & (B_SUM | B_SSD)) s += os;
          if (IS_PROD) s *= os;
          cn += ocn;
          if (OPS & B_MIN) mm = omm < mm ? omm : mm;
          if (OPS & B_MAX) mm = omm > mm ? omm : mm;
          nf |= onf;
        }
      }
      const uint32_t nlc = __shfl_down(lc, 1, 64);
      const bool boundary = (lane == 63) || (nlc != lc);
      if (boundary && lc != LC_INVALID) {
        /* s is already SumT: never narrow through Acc here — for SSD with
         * integer V, Acc is int64 and the cast would truncate the run's
         * double d^2 partial (found by the huge fuzz, seed 246810 case 21) */
        if (OPS & (B_SUM | B_SSD)) acc_add(&s_sum[lc], s);
        if (IS_PROD) acc_mul(&s_sum[lc], s);
        if ((OPS & B_CNT) && cn) atomicAdd(&s_cnt[lc], cn);
        if (OPS & B_PRESENT) s_present[lc] = 1u;
        /* gate on cn (count of non-NaN rows in the run): a real extreme can
         * encode to the combine identity (e.g. INT_MAX under min) */
        if ((OPS & B_MIN) && cn) enc_min(&s_mm[lc], mm);
        if ((OPS & B_MAX) && cn) enc_max(&s_mm[lc], mm);
        if ((OPS & B_NANFLAG) && nf) s_nanflag[lc] = 1u;
      }
    }
  }
  __syncthreads();
  for (int g = tid; g < ng_here; g += blockDim.x) {
    const int64_t o = gbase + g;
    if ((OPS & (B_SUM | B_SSD)) && s_sum[g] != (SumT)0) acc_add(&((SumT*)out_sum)[o], s_sum[g]);
    if (IS_PROD && s_sum[g] != (SumT)1) acc_mul(&((SumT*)out_sum)[o], s_sum[g]);
    if ((OPS & B_CNT) && s_cnt[g])
      atomicAdd(reinterpret_cast<unsigned long long*>(&out_count[o]), (unsigned long long)s_cnt[g]);
    if ((OPS & B_PRESENT) && s_present[g]) out_present[o] = 1u;
    if (OPS & B_MIN) enc_min(&((Enc*)out_min)[o], s_mm[g]);
    if (OPS & B_MAX) enc_max(&((Enc*)out_max)[o], s_mm[g]);
    if ((OPS & B_NANFLAG) && s_nanflag[g]) out_nanflag[o] = 1u;
  }
}

/* per-bucket LDS-binned reduce of the scattered pairs; flush with global
 * atomics (a few chunks per bucket at most) into the FINAL bins */
template <typename V, int OPS>
__launch_bounds__(BLOCK_LDS) __global__ void k_reduce_bucket(
    const PairT<V>* __restrict__ pairs, const uint32_t* __restrict__ base,
    uint32_t cap2 /* 0: base[b]..base[b+1]; else b*cap2..min(base[b],(b+1)*cap2) */,
    int64_t chunk, int gpb, int shift, int64_t ngroups,
    const double* __restrict__ means, int skipnan, BinLayout lay,
    void* out_sum, int64_t* out_count, uint32_t* out_present, void* out_min,
    void* out_max, uint32_t* out_nanflag) {
  using TR = Traits<V>;
  using Acc = typename TR::Acc;
  using SumT = typename std::conditional<
      (OPS & (B_SSD | B_WELFORD)) != 0, double,
      typename std::conditional<(OPS & (B_IDXMIN | B_IDXMAX)) != 0, int64_t, Acc>::type>::type;
  using Enc = typename TR::Enc;
  constexpr bool IS_PROD = (OPS & B_PROD) != 0;

  const int b = blockIdx.y;
  int64_t bkt_begin, bkt_end;
  if (cap2) {
    bkt_begin = (int64_t)(uint32_t)b * cap2;
    bkt_end = base[b];
    const int64_t lim = bkt_begin + cap2;
    if (bkt_end > lim) bkt_end = lim;
  } else {
    bkt_begin = base[b];
    bkt_end = base[b + 1];
  }
  const int64_t start = bkt_begin + (int64_t)blockIdx.x * chunk;
  if (start >= bkt_end) return;
  const int64_t end = (start + chunk < bkt_end) ? start + chunk : bkt_end;
  const int64_t gbase = (int64_t)b << shift;
  const int ng_here = (int)(((gbase + gpb) <= ngroups) ? gpb : (ngroups - gbase));

  extern __shared__ __attribute__((aligned(16))) char smem_rb[];
  SumT* s_sum = (OPS & (B_SUM | B_SSD | B_PROD | B_IDXMIN | B_IDXMAX))
                    ? (SumT*)(smem_rb + lay.sum_off) : nullptr;
  uint32_t* s_cnt = (OPS & B_CNT) ? (uint32_t*)(smem_rb + lay.cnt_off) : nullptr;
  uint32_t* s_present = (OPS & B_PRESENT) ? (uint32_t*)(smem_rb + lay.present_off) : nullptr;
  Enc* s_mm = (OPS & (B_MIN | B_MAX)) ? (Enc*)(smem_rb + lay.minmax_off) : nullptr;
  uint32_t* s_nanflag = (OPS & B_NANFLAG) ? (uint32_t*)(smem_rb + lay.nanflag_off) : nullptr;

  const int tid = threadIdx.x;
  for (int g = tid; g < ng_here; g += blockDim.x) {
    if (OPS & (B_SUM | B_SSD)) s_sum[g] = (SumT)0;
    if (IS_PROD) s_sum[g] = (SumT)1;
    if (OPS & B_IDXMIN) s_sum[g] = (SumT)INT64_MAX;
    if (OPS & B_IDXMAX) s_sum[g] = (SumT)(-1);
    if (OPS & B_CNT) s_cnt[g] = 0u;
    if (OPS & B_PRESENT) s_present[g] = 0u;
    if (OPS & B_MIN) s_mm[g] = (Enc)~(Enc)0;
    if (OPS & B_MAX) s_mm[g] = (Enc)0;
    if (OPS & B_NANFLAG) s_nanflag[g] = 0u;
  }
  __syncthreads();

  auto body = [&](const PairT<V> p) {
    const uint32_t lc = p.lc;
    const V v = p.v;
    const bool vnan = TR::isnan_(v);
    if (OPS & B_PRESENT) s_present[lc] = 1u;
    if (vnan && skipnan) return;
    if (OPS & B_SUM) acc_add(&s_sum[lc], (Acc)v);
    if (IS_PROD) acc_mul(&s_sum[lc], (Acc)v);
    if (OPS & B_SSD) {
      const double d = (double)v - means[gbase + lc];
      atomicAdd((double*)s_sum + lc, d * d);
    }
    if (OPS & B_CNT) {
      if (!vnan) atomicAdd(&s_cnt[lc], 1u);
    }
    if (OPS & (B_MIN | B_MAX)) {
      if (vnan) {
        if (OPS & B_NANFLAG) s_nanflag[lc] = 1u;
      } else {
        if (OPS & B_MIN) enc_min(&s_mm[lc], TR::enc(v));
        if (OPS & B_MAX) enc_max(&s_mm[lc], TR::enc(v));
      }
    }
  };
  if constexpr (sizeof(PairT<V>) == 8) {
    /* two pairs per thread per trip (one 16-B load): doubles the per-wave
     * outstanding-load count — this kernel is latency-parked, not
     * bandwidth-bound (SQ_WAIT_ANY ~88% of WAVE_CYCLES before this) */
    struct Pair2 { PairT<V> a, b; };
    const int64_t s2 = (start + 1) & ~1LL;
    if (start < s2 && start < end && tid == 0) body(pairs[start]);
    const int64_t nv = (end - s2) / 2;
    const Pair2* __restrict__ vp = (const Pair2*)(pairs + s2);
    for (int64_t j = tid; j < nv; j += blockDim.x) {
      const Pair2 q = vp[j];
      body(q.a);
      body(q.b);
    }
    const int64_t rem = s2 + nv * 2;
    if (rem < end && tid == 0) body(pairs[rem]);
  } else {
    for (int64_t i = start + tid; i < end; i += blockDim.x) body(pairs[i]);
  }
  __syncthreads();
  /* flush into the final bins (out_min/out_max hold ENCODED values until
   * k_decode, exactly like the global-atomic path) */
  for (int g = tid; g < ng_here; g += blockDim.x) {
    const int64_t o = gbase + g;
    if ((OPS & (B_SUM | B_SSD)) && s_sum[g] != (SumT)0) acc_add(&((SumT*)out_sum)[o], s_sum[g]);
    if (IS_PROD && s_sum[g] != (SumT)1) acc_mul(&((SumT*)out_sum)[o], s_sum[g]);
    if ((OPS & B_CNT) && s_cnt[g])
      atomicAdd(reinterpret_cast<unsigned long long*>(&out_count[o]), (unsigned long long)s_cnt[g]);
    if ((OPS & B_PRESENT) && s_present[g]) out_present[o] = 1u;
    if (OPS & B_MIN) enc_min(&((Enc*)out_min)[o], s_mm[g]);
    if (OPS & B_MAX) enc_max(&((Enc*)out_max)[o], s_mm[g]);
    if ((OPS & B_NANFLAG) && s_nanflag[g]) out_nanflag[o] = 1u;
  }
}


/* second bucket pass for the pair-payload arg-reductions (FH_SET_ARG*_PAIR):
 * re-reads the scattered pairs and keeps, per group, the SMALLEST row index
 * among rows matching the group's extremum (first occurrence = np.argmin /
 * np.argmax tie rule; numeric == so +-0.0 match each other exactly as the
 * reference's target-match pass does). Targets and nanflags are staged into
 * LDS once per bucket — after the partition they are bucket-local, unlike
 * the two-pass atomic form's random gathers over the full 80 MB bins. */
template <typename V>
__launch_bounds__(BLOCK_LDS) __global__ void k_reduce_bucket_argrow(
    const PairT<V>* __restrict__ pairs, const uint32_t* __restrict__ base,
    uint32_t cap2, int64_t chunk, int gpb, int shift, int64_t ngroups,
    int skipnan, const V* __restrict__ target,
    const uint32_t* __restrict__ nanflag, int64_t* __restrict__ out_idx) {
  using TR = Traits<V>;
  const int b = blockIdx.y;
  int64_t bkt_begin, bkt_end;
  if (cap2) {
    bkt_begin = (int64_t)(uint32_t)b * cap2;
    bkt_end = base[b];
    const int64_t lim = bkt_begin + cap2;
    if (bkt_end > lim) bkt_end = lim;
  } else {
    bkt_begin = base[b];
    bkt_end = base[b + 1];
  }
  const int64_t start = bkt_begin + (int64_t)blockIdx.x * chunk;
  if (start >= bkt_end) return;
  const int64_t end = (start + chunk < bkt_end) ? start + chunk : bkt_end;
  const int64_t gbase = (int64_t)b << shift;
  const int ng_here = (int)(((gbase + gpb) <= ngroups) ? gpb : (ngroups - gbase));

  extern __shared__ __attribute__((aligned(16))) char smem_ar[];
  V* s_target = (V*)smem_ar;
  uint32_t* s_row = (uint32_t*)(smem_ar + (int64_t)gpb * sizeof(V));
  uint32_t* s_nan = nanflag ? s_row + gpb : nullptr;
  const int tid = threadIdx.x;
  for (int g = tid; g < ng_here; g += blockDim.x) {
    s_target[g] = target[gbase + g];
    s_row[g] = 0xFFFFFFFFu;
    if (s_nan) s_nan[g] = nanflag[gbase + g];
  }
  __syncthreads();
  for (int64_t i = start + tid; i < end; i += blockDim.x) {
    const PairT<V> p = pairs[i];
    const uint32_t lc = p.lc;
    const V v = p.v;
    const bool vnan = TR::isnan_(v);
    bool match;
    if (skipnan)
      match = !vnan && v == s_target[lc];
    else if (s_nan && s_nan[lc])
      match = vnan; /* non-skip arg* land on the first NaN, as numpy does */
    else
      match = !vnan && v == s_target[lc];
    if (match) atomicMin(&s_row[lc], p.pad);
  }
  __syncthreads();
  for (int g = tid; g < ng_here; g += blockDim.x) {
    const uint32_t r = s_row[g];
    if (r != 0xFFFFFFFFu)
      idx_min(&out_idx[gbase + g], (int64_t)r);
  }
}

/* ---- partition-path host plumbing ---------------------------------------- */
template <typename V, int OPS>
int init_outs(fh_call* c, int64_t nbins, hipStream_t stream) {
  using TR = Traits<V>;
  if (OPS & (B_SUM | B_SSD)) FH_CHECK(hipMemsetAsync(c->out_sum, 0, nbins * 8, stream));
  if (OPS & (B_IDXMIN | B_IDXMAX)) {
    int fb = (int)((nbins + 255) / 256);
    hipLaunchKernelGGL(k_fill_i64, dim3(fb), dim3(256), 0, stream,
                       (int64_t*)c->out_sum, nbins,
                       (OPS & B_IDXMIN) ? INT64_MAX : (int64_t)-1);
    FH_CHECK(hipGetLastError());
  }
  if (OPS & B_ARGROW) {
    /* row-index bins (out_sum as int64): min-row on both argmin and argmax
     * (first occurrence), sentinel INT64_MAX for empty */
    int fb = (int)((nbins + 255) / 256);
    hipLaunchKernelGGL(k_fill_i64, dim3(fb), dim3(256), 0, stream,
                       (int64_t*)c->out_sum, nbins, INT64_MAX);
    FH_CHECK(hipGetLastError());
  }
  if (OPS & B_PROD) {
    int fb = (int)((nbins + 255) / 256);
    if (c->vdtype == FH_F32 || c->vdtype == FH_F64)
      hipLaunchKernelGGL(k_fill_f64, dim3(fb), dim3(256), 0, stream,
                         (double*)c->out_sum, nbins, 1.0);
    else
      hipLaunchKernelGGL(k_fill_i64, dim3(fb), dim3(256), 0, stream,
                         (int64_t*)c->out_sum, nbins, (int64_t)1);
    FH_CHECK(hipGetLastError());
  }
  if (OPS & B_CNT) FH_CHECK(hipMemsetAsync(c->out_count, 0, nbins * 8, stream));
  if (OPS & B_PRESENT) FH_CHECK(hipMemsetAsync(c->out_present, 0, nbins * 4, stream));
  if (OPS & B_MIN)
    FH_CHECK(hipMemsetAsync(c->out_min, 0xFF, nbins * sizeof(typename TR::Enc), stream));
  if (OPS & B_MAX)
    FH_CHECK(hipMemsetAsync(c->out_max, 0x00, nbins * sizeof(typename TR::Enc), stream));
  if (OPS & B_NANFLAG) FH_CHECK(hipMemsetAsync(c->out_nanflag, 0, nbins * 4, stream));
  return 0;
}

/* argrow phase launcher shared by the three partition variants */
template <typename V, int OPS, typename PP>
int launch_argrow_phase(fh_call* c, const PP& pp, const PairT<V>* pairs,
                        const uint32_t* ends, uint32_t capF, int64_t maxspan,
                        hipStream_t stream) {
  if constexpr ((OPS & B_ARGROW) != 0 && sizeof(PairT<V>) == 16) {
    const int skipnan = (c->flags & FH_SKIPNAN) ? 1 : 0;
    const V* target = (const V*)((OPS & B_MIN) ? c->out_min : c->out_max);
    const uint32_t* nan = skipnan ? nullptr : (const uint32_t*)c->out_nanflag;
    const int64_t chunk = 1 << 19;
    int maxchunks = (int)((maxspan + chunk - 1) / chunk);
    if (maxchunks < 1) maxchunks = 1;
    const int64_t lds = ((int64_t)pp.gpb * (sizeof(V) + 8) + 255) / 256 * 256;
    auto kern = k_reduce_bucket_argrow<V>;
    FH_CHECK(hipFuncSetAttribute((const void*)kern,
                                 hipFuncAttributeMaxDynamicSharedMemorySize,
                                 (int)lds));
    hipLaunchKernelGGL(kern, dim3(maxchunks, pp.B), dim3(BLOCK_LDS), lds,
                       stream, pairs, ends, capF, chunk, pp.gpb, pp.shift,
                       c->ngroups, skipnan, target, nan, (int64_t*)c->out_sum);
    FH_CHECK(hipGetLastError());
  }
  return 0;
}

struct PartPlan {
  int shift, gpb, B, Bpad;
  bool two_level;      /* B > 64: scatter via <= 64 super-buckets first */
  int B1;              /* super-bucket count (two_level) */
  BinLayout lay;       /* per-bucket bins (gpb entries) */
  int64_t pairs_off, pairs2_off, counts_off, base_off, baseA_off, cursors_off, bytes;
  int64_t overflow_off;
  uint32_t cap1, cap2; /* optimistic region strides (super / fine) */
  bool optimistic;     /* capacity regions fit u32 cursors */
  int64_t scatter_lds, scatter2_lds;
  bool feasible;
};

/* FH_PART_MODE: 0 auto (one-level direct scatter once two levels would be
 * needed), 1 force the two-level staged path, 2 force direct everywhere.
 * FH_PART_BINKB: pass-C LDS bin budget in KiB (default 150 of the CU's
 * 160; larger bins -> fewer, bigger buckets -> fewer reservation atomics). */
inline int fh_part_mode() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("FH_PART_MODE");
    v = e ? atoi(e) : 0;
    if (v < 0 || v > 2) v = 0;
  }
  return v;
}

/* FH_PART_TILE: 0 (default) = full scatter staging tile (f32 6144 / f64
 * 3072 pairs, ~76 KB LDS, 2 blocks/CU); 1 = small tile (f32 4096 / f64
 * 2048, ~51/43 KB, 3 blocks/CU) — occupancy-vs-coalescing A/B knob. */
inline int fh_part_tile() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("FH_PART_TILE");
    v = (e && atoi(e) != 0) ? 1 : 0;
  }
  return v;
}

constexpr int64_t part_tile_lds(int T, int pairsz) {
  return (int64_t)(PART_SUB * PART_NW + PART_NW + 1 + PART_SUB + 3) * 4 +
         (int64_t)T * 4 + (int64_t)T * pairsz;
}

inline int64_t fh_part_bin_bytes() {
  static int64_t v = -1;
  if (v < 0) {
    /* default 120 measured best: for 16 B/group sets it keeps shift 12
     * (4096 groups/bucket, 2 reduce WGs/CU) — 10.6 ms vs 11.4 at 150 KiB
     * (shift 13, 1 WG/CU) on the 1e9-row/1e7-group sum (r02 A/B) */
    const char* e = getenv("FH_PART_BINKB");
    int kb = e ? atoi(e) : 120;
    if (kb < 16) kb = 16;
    v = (int64_t)kb * 1024;
    if (v > LDS_MAX) v = LDS_MAX;
  }
  return v;
}

template <typename V>
PartPlan part_plan(const fh_call* c) {
  PartPlan p{};
  p.feasible = false;
  if (c->n >= ((int64_t)1 << 31) || c->ngroups <= 0) return p;
  const int bits = set_bits(c->op_set);
  if (bits & (B_IDXMIN | B_IDXMAX)) return p; /* pairs carry no row index */
  if ((bits & B_ARGROW) &&
      (sizeof(PairT<V>) != 16 || c->n + c->row_offset >= ((int64_t)1 << 32)))
    return p; /* pad word (8-byte dtypes) carries the 32-bit row */
  int shift = 13;
  while (shift > 8 && bin_layout<V>(bits, (int64_t)1 << shift, 4).bytes > fh_part_bin_bytes())
    shift--;
  BinLayout lay = bin_layout<V>(bits, (int64_t)1 << shift, 4);
  if (lay.bytes > fh_part_bin_bytes()) return p;
  int64_t B64 = (c->ngroups + ((int64_t)1 << shift) - 1) >> shift;
  if (B64 > 4096) return p;
  p.B = (int)B64;
  p.shift = shift;
  p.gpb = 1 << shift;
  p.lay = lay;
  p.two_level = p.B > PART_SUB;
  p.B1 = p.two_level ? (p.B + PART_SUB - 1) / PART_SUB : 0;
  /* pass-A bucket count: B1 when two-level, else the fine B */
  const int bA = p.two_level ? p.B1 : p.B;
  if (bA > PART_SUB) return p; /* scatter passes handle <= 64 buckets */
  p.Bpad = PART_SUB;
  constexpr int T = sizeof(V) == 4 ? 6144 : 3072;
  const int64_t tile_lds = (int64_t)(PART_SUB * PART_NW + PART_NW + 1 + PART_SUB + 3) * 4 +
                           (int64_t)T * 4 + (int64_t)T * sizeof(PairT<V>);
  p.scatter_lds = tile_lds;
  p.scatter2_lds = tile_lds;
  if (p.scatter_lds > LDS_MAX) return p;
  /* optimistic capacity regions: uniform 25% + 8K rows of slack per bucket;
   * overflow (extreme skew) falls back to the exact counted path */
  auto cap_of = [&](int nb) {
    const int64_t mean = (c->n + nb - 1) / nb;
    return (uint32_t)(((mean + (mean >> 2) + 8192) + 63) / 64 * 64);
  };
  p.cap2 = cap_of(p.B);
  p.cap1 = p.two_level ? cap_of(p.B1) : 0;
  const int64_t reg2 = (int64_t)p.B * p.cap2;
  const int64_t reg1 = p.two_level ? (int64_t)p.B1 * p.cap1 : reg2;
  p.optimistic = reg1 < ((int64_t)1 << 31) && reg2 < ((int64_t)1 << 31);
  int64_t off = 0;
  auto carve = [&](int64_t b) {
    int64_t o = off;
    off += ((b + 255) / 256) * 256;
    return o;
  };
  const int64_t p1elems = p.optimistic ? std::max(c->n, reg1) : c->n;
  const int64_t p2elems = p.optimistic ? std::max(c->n, reg2) : c->n;
  p.pairs_off = carve(p1elems * (int64_t)sizeof(PairT<V>));
  p.pairs2_off = p.two_level ? carve(p2elems * (int64_t)sizeof(PairT<V>)) : p.pairs_off;
  p.counts_off = carve((int64_t)p.B * 4);
  p.base_off = carve(((int64_t)p.B + 1) * 4);
  p.baseA_off = carve(((int64_t)p.B1 + 1) * 4);
  p.cursors_off = carve((int64_t)p.B * 4);
  p.overflow_off = carve(256);
  p.bytes = off;
  p.feasible = true;
  return p;
}

template <typename V, typename L, int OPS>
int launch_partition_exact(fh_call* c, const PartPlan& pp) {
  hipStream_t stream = (hipStream_t)c->stream;
  const int skipnan = (c->flags & FH_SKIPNAN) ? 1 : 0;
  constexpr bool CROW = (OPS & B_ARGROW) != 0 && sizeof(PairT<V>) == 16;
  char* scr = (char*)c->scratch;
  uint32_t* overflow = (uint32_t*)(scr + pp.overflow_off);
  uint32_t* counts = (uint32_t*)(scr + pp.counts_off);
  uint32_t* based = (uint32_t*)(scr + pp.base_off);
  uint32_t* baseAd = (uint32_t*)(scr + pp.baseA_off);
  uint32_t* cursors = (uint32_t*)(scr + pp.cursors_off);
  PairT<V>* pairs = (PairT<V>*)(scr + pp.pairs_off);
  PairT<V>* pairs2 = (PairT<V>*)(scr + pp.pairs2_off);

  FH_CHECK(hipMemsetAsync(counts, 0, (int64_t)pp.B * 4, stream));
  {
    int64_t wb = (c->n + 255) / 256;
    int nb0 = (int)(wb < 2048 ? (wb > 0 ? wb : 1) : 2048);
    /* the count pass histograms the FINE buckets (up to 4096) */
    const int64_t hist_lds = ((int64_t)pp.B + 63) / 64 * 64 * 4;
    hipLaunchKernelGGL((k_part_count<L>), dim3(nb0), dim3(256), hist_lds,
                       stream, (const L*)c->labels, (const L*)c->labels2, c->n,
                       c->ngroups, c->g0, c->g1, pp.shift, pp.B, counts);
    FH_CHECK(hipGetLastError());
  }
  /* bucket directory on the host (one sync; B <= 4096 words) */
  static thread_local uint32_t h_counts[4096], h_base[4097];
  FH_CHECK(hipMemcpyAsync(h_counts, counts, (int64_t)pp.B * 4,
                          hipMemcpyDeviceToHost, stream));
  FH_CHECK(hipStreamSynchronize(stream));
  uint32_t acc = 0, maxc = 0;
  for (int i = 0; i < pp.B; ++i) {
    h_base[i] = acc;
    acc += h_counts[i];
    if (h_counts[i] > maxc) maxc = h_counts[i];
  }
  h_base[pp.B] = acc;
  FH_CHECK(hipMemcpyAsync(based, h_base, ((int64_t)pp.B + 1) * 4,
                          hipMemcpyHostToDevice, stream));
  /* super-bucket directory: sums of PART_SUB fine buckets */
  static thread_local uint32_t h_baseA[4097];
  uint32_t maxA = 0;
  if (pp.two_level) {
    for (int sbi = 0; sbi <= pp.B1; ++sbi) {
      const int fb = sbi * PART_SUB;
      h_baseA[sbi] = h_base[fb < pp.B ? fb : pp.B];
    }
    for (int sbi = 0; sbi < pp.B1; ++sbi) {
      const uint32_t cnt = h_baseA[sbi + 1] - h_baseA[sbi];
      if (cnt > maxA) maxA = cnt;
    }
    FH_CHECK(hipMemcpyAsync(baseAd, h_baseA, ((int64_t)pp.B1 + 1) * 4,
                            hipMemcpyHostToDevice, stream));
    /* pass-A cursors: super-bucket bases (first B1 words of cursors) */
    FH_CHECK(hipMemcpyAsync(cursors, h_baseA, (int64_t)pp.B1 * 4,
                            hipMemcpyHostToDevice, stream));
  } else {
    FH_CHECK(hipMemcpyAsync(cursors, h_base, (int64_t)pp.B * 4,
                            hipMemcpyHostToDevice, stream));
  }

  {
    const int shA = pp.two_level ? pp.shift + 6 : pp.shift;
    const int bA = pp.two_level ? pp.B1 : pp.B;
    auto launch1 = [&](auto tc) {
      constexpr int T = decltype(tc)::value;
      auto kern = k_part_scatter<V, L, CROW, T>;
      const int64_t lds = part_tile_lds(T, (int)sizeof(PairT<V>));
      hipError_t e = hipFuncSetAttribute(
          (const void*)kern, hipFuncAttributeMaxDynamicSharedMemorySize,
          (int)lds);
      if (e != hipSuccess) return e;
      int64_t wb = (c->n + T - 1) / T;
      int nb1 = (int)(wb < 1024 ? (wb > 0 ? wb : 1) : 1024);
      hipLaunchKernelGGL(kern, dim3(nb1), dim3(PART_BLOCK), lds, stream,
                         (const V*)c->values, (const L*)c->labels,
                         (const L*)c->labels2, c->n, c->ngroups, c->g0, c->g1,
                         shA, bA, pp.Bpad, cursors, 0u, overflow, pairs,
                         c->row_offset);
      return hipGetLastError();
    };
    hipError_t e1;
    if (fh_part_tile())
      e1 = launch1(std::integral_constant<int, (sizeof(V) == 4 ? 4096 : 2048)>{});
    else
      e1 = launch1(std::integral_constant<int, (sizeof(V) == 4 ? 6144 : 3072)>{});
    FH_CHECK(e1);
  }
  if (pp.two_level) {
    /* fine-bucket cursors, then the in-super-bucket scatter */
    FH_CHECK(hipMemcpyAsync(cursors, h_base, (int64_t)pp.B * 4,
                            hipMemcpyHostToDevice, stream));
    auto launch2 = [&](auto tc) {
      constexpr int T = decltype(tc)::value;
      auto kern2 = k_part_scatter2<V, CROW, T>;
      const int64_t lds = part_tile_lds(T, (int)sizeof(PairT<V>));
      hipError_t e = hipFuncSetAttribute(
          (const void*)kern2, hipFuncAttributeMaxDynamicSharedMemorySize,
          (int)lds);
      if (e != hipSuccess) return e;
      int tiles_x = (int)((maxA + T - 1) / T);
      if (tiles_x > 64) tiles_x = 64;
      if (tiles_x < 1) tiles_x = 1;
      hipLaunchKernelGGL(kern2, dim3(tiles_x, pp.B1), dim3(PART_BLOCK), lds,
                         stream, pairs, baseAd, 0u, pp.shift,
                         pp.B, cursors, 0u, overflow, pairs2);
      return hipGetLastError();
    };
    hipError_t e2;
    if (fh_part_tile())
      e2 = launch2(std::integral_constant<int, (sizeof(V) == 4 ? 4096 : 2048)>{});
    else
      e2 = launch2(std::integral_constant<int, (sizeof(V) == 4 ? 6144 : 3072)>{});
    FH_CHECK(e2);
  }
  {
    int rc = init_outs<V, OPS>(c, c->ngroups, stream);
    if (rc) return rc;
  }
  {
    auto kern = k_reduce_bucket<V, OPS>;
    FH_CHECK(hipFuncSetAttribute((const void*)kern,
                                 hipFuncAttributeMaxDynamicSharedMemorySize,
                                 (int)pp.lay.bytes));
    const int64_t chunk = 1 << 19;
    int maxchunks = (int)((maxc + chunk - 1) / chunk);
    if (maxchunks < 1) maxchunks = 1;
    hipLaunchKernelGGL(kern, dim3(maxchunks, pp.B), dim3(BLOCK_LDS),
                       pp.lay.bytes, stream, pp.two_level ? pairs2 : pairs,
                       based, 0u, chunk, pp.gpb, pp.shift, c->ngroups, c->means,
                       skipnan, pp.lay, c->out_sum, c->out_count,
                       c->out_present, c->out_min, c->out_max, c->out_nanflag);
    FH_CHECK(hipGetLastError());
  }
  if (OPS & (B_MIN | B_MAX)) {
    int db = (int)((c->ngroups + 255) / 256);
    hipLaunchKernelGGL((k_decode<V, OPS>), dim3(db), dim3(256), 0, stream,
                       c->ngroups, c->out_min, c->out_max, c->out_count,
                       c->out_present);
    FH_CHECK(hipGetLastError());
  }
  {
    int rc = launch_argrow_phase<V, OPS>(c, pp, pp.two_level ? pairs2 : pairs,
                                         based, 0u, (int64_t)maxc, stream);
    if (rc) return rc;
  }
  c->path_used = 4;
  return 0;
}

/* one-level direct partition (optimistic): scatter straight into the
 * fine-bucket capacity regions with k_part_scatter_direct — no second
 * scatter pass and no LDS staging — then the per-bucket LDS-binned
 * reduce. ~28 B/row streamed vs the two-level's ~44. Overflow falls back
 * to the exact two-level path (the scratch keeps its layout). */
template <typename V, typename L, int OPS>
int launch_partition_direct(fh_call* c, const PartPlan& pp) {
  hipStream_t stream = (hipStream_t)c->stream;
  const int skipnan = (c->flags & FH_SKIPNAN) ? 1 : 0;
  constexpr bool CROW = (OPS & B_ARGROW) != 0 && sizeof(PairT<V>) == 16;
  char* scr = (char*)c->scratch;
  uint32_t* cursors = (uint32_t*)(scr + pp.cursors_off);
  uint32_t* overflow = (uint32_t*)(scr + pp.overflow_off);
  PairT<V>* pairs2 = (PairT<V>*)(scr + pp.pairs2_off);

  FH_CHECK(hipMemsetAsync(overflow, 0, 4, stream));
  hipLaunchKernelGGL(k_init_cursors, dim3((pp.B + 255) / 256), dim3(256), 0,
                     stream, cursors, pp.B, pp.cap2);
  FH_CHECK(hipGetLastError());
  {
    auto kern = k_part_scatter_direct<V, L, CROW>;
    const int64_t hist_lds = ((int64_t)pp.B + 63) / 64 * 64 * 4;
    FH_CHECK(hipFuncSetAttribute((const void*)kern,
                                 hipFuncAttributeMaxDynamicSharedMemorySize,
                                 (int)hist_lds));
    constexpr int T = PBD * (sizeof(V) == 4 ? 24 : 12);
    int64_t wb = (c->n + T - 1) / T;
    int nb = (int)(wb < 1536 ? (wb > 0 ? wb : 1) : 1536);
    hipLaunchKernelGGL(kern, dim3(nb), dim3(PBD), hist_lds, stream,
                       (const V*)c->values, (const L*)c->labels,
                       (const L*)c->labels2, c->n, c->ngroups, c->g0, c->g1,
                       pp.shift, pp.B, cursors, pp.cap2, overflow, pairs2,
                       c->row_offset);
    FH_CHECK(hipGetLastError());
  }
  {
    int rc = init_outs<V, OPS>(c, c->ngroups, stream);
    if (rc) return rc;
  }
  {
    auto kern = k_reduce_bucket<V, OPS>;
    FH_CHECK(hipFuncSetAttribute((const void*)kern,
                                 hipFuncAttributeMaxDynamicSharedMemorySize,
                                 (int)pp.lay.bytes));
    const int64_t chunk = 1 << 19;
    int maxchunks = (int)(((int64_t)pp.cap2 + chunk - 1) / chunk);
    if (maxchunks < 1) maxchunks = 1;
    hipLaunchKernelGGL(kern, dim3(maxchunks, pp.B), dim3(BLOCK_LDS),
                       pp.lay.bytes, stream, pairs2, cursors, pp.cap2, chunk,
                       pp.gpb, pp.shift, c->ngroups, c->means, skipnan, pp.lay,
                       c->out_sum, c->out_count, c->out_present, c->out_min,
                       c->out_max, c->out_nanflag);
    FH_CHECK(hipGetLastError());
  }
  if (OPS & (B_MIN | B_MAX)) {
    int db = (int)((c->ngroups + 255) / 256);
    hipLaunchKernelGGL((k_decode<V, OPS>), dim3(db), dim3(256), 0, stream,
                       c->ngroups, c->out_min, c->out_max, c->out_count,
                       c->out_present);
    FH_CHECK(hipGetLastError());
  }
  {
    int rc = launch_argrow_phase<V, OPS>(c, pp, pairs2, cursors, pp.cap2,
                                         (int64_t)pp.cap2, stream);
    if (rc) return rc;
  }
  uint32_t h_ov = 0;
  FH_CHECK(hipMemcpyAsync(&h_ov, overflow, 4, hipMemcpyDeviceToHost, stream));
  FH_CHECK(hipStreamSynchronize(stream));
  if (h_ov) return launch_partition_exact<V, L, OPS>(c, pp);
  c->path_used = 6;
  return 0;
}

/* optimistic partition: capacity regions replace the counting pre-pass and
 * every host round trip except the final 4-byte overflow check */
template <typename V, typename L, int OPS>
int launch_partition(fh_call* c, const PartPlan& pp) {
  if (!pp.optimistic) return launch_partition_exact<V, L, OPS>(c, pp);
  constexpr bool CROW = (OPS & B_ARGROW) != 0 && sizeof(PairT<V>) == 16;
  /* The one-level direct variant is kept for evidence (FH_PART_MODE=2) but
   * is NOT the default: measured on MI355X (profiles/r02_direct_pmc.md) its
   * partial-line scatter stores amplify 3.5x (WRITE_SIZE 28.1 GB for 8 GB
   * of pairs) and the ~1e8 reservation atomics add fabric RMW, so its
   * effective traffic exceeds the staged two-level's 44 GB/row-stream. */
  if (fh_part_mode() == 2) return launch_partition_direct<V, L, OPS>(c, pp);
  hipStream_t stream = (hipStream_t)c->stream;
  const int skipnan = (c->flags & FH_SKIPNAN) ? 1 : 0;
  char* scr = (char*)c->scratch;
  uint32_t* cursorsA = (uint32_t*)(scr + pp.counts_off); /* reused slot */
  uint32_t* cursors = (uint32_t*)(scr + pp.cursors_off);
  uint32_t* overflow = (uint32_t*)(scr + pp.overflow_off);
  PairT<V>* pairs = (PairT<V>*)(scr + pp.pairs_off);
  PairT<V>* pairs2 = (PairT<V>*)(scr + pp.pairs2_off);

  FH_CHECK(hipMemsetAsync(overflow, 0, 4, stream));
  const int bA = pp.two_level ? pp.B1 : pp.B;
  const uint32_t capA = pp.two_level ? pp.cap1 : pp.cap2;
  hipLaunchKernelGGL(k_init_cursors, dim3(1), dim3(PART_SUB), 0, stream,
                     cursorsA, bA, capA);
  FH_CHECK(hipGetLastError());
  {
    const int shA = pp.two_level ? pp.shift + 6 : pp.shift;
    auto launch1 = [&](auto tc) {
      constexpr int T = decltype(tc)::value;
      auto kern = k_part_scatter<V, L, CROW, T>;
      const int64_t lds = part_tile_lds(T, (int)sizeof(PairT<V>));
      hipError_t e = hipFuncSetAttribute(
          (const void*)kern, hipFuncAttributeMaxDynamicSharedMemorySize,
          (int)lds);
      if (e != hipSuccess) return e;
      int64_t wb = (c->n + T - 1) / T;
      int nb1 = (int)(wb < 1024 ? (wb > 0 ? wb : 1) : 1024);
      hipLaunchKernelGGL(kern, dim3(nb1), dim3(PART_BLOCK), lds, stream,
                         (const V*)c->values, (const L*)c->labels,
                         (const L*)c->labels2, c->n, c->ngroups, c->g0, c->g1,
                         shA, bA, pp.Bpad, cursorsA, capA, overflow, pairs,
                         c->row_offset);
      return hipGetLastError();
    };
    hipError_t e1;
    if (fh_part_tile())
      e1 = launch1(std::integral_constant<int, (sizeof(V) == 4 ? 4096 : 2048)>{});
    else
      e1 = launch1(std::integral_constant<int, (sizeof(V) == 4 ? 6144 : 3072)>{});
    FH_CHECK(e1);
  }
  if (pp.two_level) {
    hipLaunchKernelGGL(k_init_cursors, dim3((pp.B + 255) / 256), dim3(256), 0,
                       stream, cursors, pp.B, pp.cap2);
    FH_CHECK(hipGetLastError());
    auto launch2 = [&](auto tc) {
      constexpr int T = decltype(tc)::value;
      auto kern2 = k_part_scatter2<V, CROW, T>;
      const int64_t lds = part_tile_lds(T, (int)sizeof(PairT<V>));
      hipError_t e = hipFuncSetAttribute(
          (const void*)kern2, hipFuncAttributeMaxDynamicSharedMemorySize,
          (int)lds);
      if (e != hipSuccess) return e;
      int tiles_x = (int)(((int64_t)pp.cap1 + T - 1) / T);
      if (tiles_x > 64) tiles_x = 64;
      if (tiles_x < 1) tiles_x = 1;
      hipLaunchKernelGGL(kern2, dim3(tiles_x, pp.B1), dim3(PART_BLOCK), lds,
                         stream, pairs, cursorsA, pp.cap1,
                         pp.shift, pp.B, cursors, pp.cap2, overflow, pairs2);
      return hipGetLastError();
    };
    hipError_t e2;
    if (fh_part_tile())
      e2 = launch2(std::integral_constant<int, (sizeof(V) == 4 ? 4096 : 2048)>{});
    else
      e2 = launch2(std::integral_constant<int, (sizeof(V) == 4 ? 6144 : 3072)>{});
    FH_CHECK(e2);
  }
  {
    int rc = init_outs<V, OPS>(c, c->ngroups, stream);
    if (rc) return rc;
  }
  {
    auto kern = k_reduce_bucket<V, OPS>;
    FH_CHECK(hipFuncSetAttribute((const void*)kern,
                                 hipFuncAttributeMaxDynamicSharedMemorySize,
                                 (int)pp.lay.bytes));
    const int64_t chunk = 1 << 19;
    const uint32_t capF = pp.cap2;
    int maxchunks = (int)(((int64_t)capF + chunk - 1) / chunk);
    if (maxchunks < 1) maxchunks = 1;
    hipLaunchKernelGGL(kern, dim3(maxchunks, pp.B), dim3(BLOCK_LDS),
                       pp.lay.bytes, stream,
                       pp.two_level ? pairs2 : pairs,
                       pp.two_level ? cursors : cursorsA, capF, chunk, pp.gpb,
                       pp.shift, c->ngroups, c->means, skipnan, pp.lay,
                       c->out_sum, c->out_count, c->out_present, c->out_min,
                       c->out_max, c->out_nanflag);
    FH_CHECK(hipGetLastError());
  }
  if (OPS & (B_MIN | B_MAX)) {
    int db = (int)((c->ngroups + 255) / 256);
    hipLaunchKernelGGL((k_decode<V, OPS>), dim3(db), dim3(256), 0, stream,
                       c->ngroups, c->out_min, c->out_max, c->out_count,
                       c->out_present);
    FH_CHECK(hipGetLastError());
  }
  {
    int rc = launch_argrow_phase<V, OPS>(
        c, pp, pp.two_level ? pairs2 : pairs,
        pp.two_level ? cursors : cursorsA, pp.cap2, (int64_t)pp.cap2, stream);
    if (rc) return rc;
  }
  uint32_t h_ov = 0;
  FH_CHECK(hipMemcpyAsync(&h_ov, overflow, 4, hipMemcpyDeviceToHost, stream));
  FH_CHECK(hipStreamSynchronize(stream));
  if (h_ov) return launch_partition_exact<V, L, OPS>(c, pp);
  c->path_used = 4;
  return 0;
}

/* ---- host-side dispatch -------------------------------------------------- */

template <typename V, typename L, int OPS>
int launch_typed(fh_call* c) {
  using TR = Traits<V>;
  hipStream_t stream = (hipStream_t)c->stream;
  const bool skipnan = (c->flags & FH_SKIPNAN) != 0;
  BinLayout lay = bin_layout<V>(OPS, c->ngroups, 4);

  if (OPS & B_ARGROW) {
    /* pair-payload arg-reductions exist only on the partition path (which
     * host-syncs, so also refused during capture — callers use the
     * two-pass form there) */
    if (c->flags & (FH_FORCE_LDS | FH_FORCE_ATOMIC | FH_NO_HOST_SYNC))
      return 11;
    PartPlan pp = part_plan<V>(c);
    if (!pp.feasible || c->scratch_bytes < pp.bytes) return 11;
    return launch_partition<V, L, OPS>(c, pp);
  }

  bool use_lds = lay.bytes <= LDS_MAX;
  if (c->flags & FH_FORCE_ATOMIC) use_lds = false;
  if (c->flags & FH_FORCE_LDS) {
    if (lay.bytes > LDS_MAX) return 2; /* cannot honor */
    use_lds = true;
  }

  if (use_lds) {
    int blocks_per_cu = lay.bytes * 2 <= LDS_MAX ? 2 : 1;
    int nblocks = NUM_CU * blocks_per_cu;
    /* do not launch more blocks than there is work or scratch for */
    int64_t work_blocks = (c->n + BLOCK_LDS - 1) / BLOCK_LDS;
    if (work_blocks < nblocks) nblocks = (int)(work_blocks > 0 ? work_blocks : 1);
    if ((int64_t)nblocks * lay.bytes > c->scratch_bytes) return 3;

    auto kern = k_reduce_lds<V, L, OPS>;
    FH_CHECK(hipFuncSetAttribute((const void*)kern,
                                 hipFuncAttributeMaxDynamicSharedMemorySize,
                                 (int)lay.bytes));
    hipLaunchKernelGGL(kern, dim3(nblocks), dim3(BLOCK_LDS), lay.bytes, stream,
                       (const V*)c->values, (const L*)c->labels,
                       (const L*)c->labels2, c->n, c->ngroups, c->g0, c->g1,
                       c->means, (const V*)c->target, c->row_offset,
                       (int)skipnan, (char*)c->scratch, lay);
    FH_CHECK(hipGetLastError());
    int cb = (int)((c->ngroups + 255) / 256);
    /* fill the chip even for tiny group counts: split the slab-block fold */
    int S = 1;
    if (c->ngroups < 65536) {
      S = (int)((524288 + c->ngroups - 1) / c->ngroups);
      if (S > nblocks) S = nblocks;
      if (S < 1) S = 1;
    }
    if (S > 1) {
      int rc = init_outs<V, OPS>(c, c->ngroups, stream);
      if (rc) return rc;
    }
    hipLaunchKernelGGL((k_combine<V, OPS>), dim3(cb, S), dim3(256), 0, stream,
                       (const char*)c->scratch, nblocks, c->ngroups, lay,
                       c->out_sum, c->out_count, c->out_present, c->out_min,
                       c->out_max, c->out_nanflag);
    FH_CHECK(hipGetLastError());
    if (S > 1 && (OPS & (B_MIN | B_MAX))) {
      hipLaunchKernelGGL((k_decode<V, OPS>), dim3(cb), dim3(256), 0, stream,
                         c->ngroups, c->out_min, c->out_max, c->out_count,
                         c->out_present);
      FH_CHECK(hipGetLastError());
    }
    c->path_used = 1;
    return 0;
  }

  /* sorted-labels direct path: skip every scatter pass — each bucket's rows
   * are already a contiguous range of the original arrays */
  if ((c->flags & FH_SORTED_LABELS) && !c->labels2 &&
      !(c->flags & FH_FORCE_ATOMIC) && !(OPS & (B_IDXMIN | B_IDXMAX))) {
    PartPlan pp = part_plan<V>(c);
    if (pp.feasible && c->scratch_bytes >= (int64_t)(pp.B + 1) * 4) {
      uint32_t* base = (uint32_t*)c->scratch;
      const int bb = (pp.B + 256) / 256;
      hipLaunchKernelGGL((k_bucket_bounds<L>), dim3(bb), dim3(256), 0, stream,
                         (const L*)c->labels, c->n, pp.B, pp.shift, base);
      FH_CHECK(hipGetLastError());
      int rc = init_outs<V, OPS>(c, c->ngroups, stream);
      if (rc) return rc;
      auto kern = k_reduce_bucket_direct<V, L, OPS>;
      FH_CHECK(hipFuncSetAttribute((const void*)kern,
                                   hipFuncAttributeMaxDynamicSharedMemorySize,
                                   (int)pp.lay.bytes));
      const int64_t chunk = 1 << 19;
      int nchunks = (int)((c->n / pp.B + chunk) / chunk) + 1;
      if (nchunks > 16) nchunks = 16;
      hipLaunchKernelGGL(kern, dim3(nchunks, pp.B), dim3(BLOCK_LDS),
                         pp.lay.bytes, stream, (const V*)c->values,
                         (const L*)c->labels, base, chunk, pp.gpb, pp.shift,
                         c->ngroups, c->means, skipnan, pp.lay, c->out_sum,
                         c->out_count, c->out_present, c->out_min, c->out_max,
                         c->out_nanflag);
      FH_CHECK(hipGetLastError());
      if (OPS & (B_MIN | B_MAX)) {
        int db = (int)((c->ngroups + 255) / 256);
        hipLaunchKernelGGL((k_decode<V, OPS>), dim3(db), dim3(256), 0, stream,
                           c->ngroups, c->out_min, c->out_max, c->out_count,
                           c->out_present);
        FH_CHECK(hipGetLastError());
      }
      c->path_used = 5;
      return 0;
    }
  }

  /* huge group counts: bucket-partition path when scratch allows. Not
   * during hipGraph capture (FH_NO_HOST_SYNC): the optimistic overflow
   * check and the exact path's counting pre-pass read back to the host
   * and synchronize the stream, which invalidates a capture. */
  if (!(c->flags & (FH_FORCE_ATOMIC | FH_NO_HOST_SYNC))) {
    PartPlan pp = part_plan<V>(c);
    if (pp.feasible && c->scratch_bytes >= pp.bytes)
      return launch_partition<V, L, OPS>(c, pp);
  }

  /* global-atomic path: memset-style init of the bins */
  const int64_t ng = c->ngroups;
  if (OPS & (B_SUM | B_SSD))
    FH_CHECK(hipMemsetAsync(c->out_sum, 0, ng * 8, stream));
  if (OPS & (B_IDXMIN | B_IDXMAX)) {
    int fb = (int)((ng + 255) / 256);
    hipLaunchKernelGGL(k_fill_i64, dim3(fb), dim3(256), 0, stream,
                       (int64_t*)c->out_sum, ng,
                       (OPS & B_IDXMIN) ? INT64_MAX : (int64_t)-1);
    FH_CHECK(hipGetLastError());
  }
  if (OPS & B_PROD) {
    int fb = (int)((ng + 255) / 256);
    if (c->vdtype == FH_F32 || c->vdtype == FH_F64)
      hipLaunchKernelGGL(k_fill_f64, dim3(fb), dim3(256), 0, stream,
                         (double*)c->out_sum, ng, 1.0);
    else
      hipLaunchKernelGGL(k_fill_i64, dim3(fb), dim3(256), 0, stream,
                         (int64_t*)c->out_sum, ng, (int64_t)1);
    FH_CHECK(hipGetLastError());
  }
  if (OPS & B_CNT) FH_CHECK(hipMemsetAsync(c->out_count, 0, ng * 8, stream));
  if (OPS & B_PRESENT) FH_CHECK(hipMemsetAsync(c->out_present, 0, ng * 4, stream));
  if (OPS & B_MIN)
    FH_CHECK(hipMemsetAsync(c->out_min, 0xFF, ng * sizeof(typename TR::Enc), stream));
  if (OPS & B_MAX)
    FH_CHECK(hipMemsetAsync(c->out_max, 0x00, ng * sizeof(typename TR::Enc), stream));
  if (OPS & B_NANFLAG) FH_CHECK(hipMemsetAsync(c->out_nanflag, 0, ng * 4, stream));

  int64_t work_blocks = (c->n / TR::VEC + BLOCK_ATOMIC - 1) / BLOCK_ATOMIC;
  int nblocks = (int)(work_blocks < 2048 ? (work_blocks > 0 ? work_blocks : 1) : 2048);
  hipLaunchKernelGGL((k_reduce_atomic<V, L, OPS>), dim3(nblocks),
                     dim3(BLOCK_ATOMIC), 0, stream, (const V*)c->values,
                     (const L*)c->labels, (const L*)c->labels2, c->n, c->ngroups,
                     c->g0, c->g1, c->means, (const V*)c->target, c->row_offset,
                     (int)skipnan, c->out_sum,
                     c->out_count, c->out_present, c->out_min, c->out_max,
                     c->out_nanflag);
  FH_CHECK(hipGetLastError());
  if (OPS & (B_MIN | B_MAX)) {
    int db = (int)((ng + 255) / 256);
    hipLaunchKernelGGL((k_decode<V, OPS>), dim3(db), dim3(256), 0, stream, ng,
                       c->out_min, c->out_max, c->out_count, c->out_present);
    FH_CHECK(hipGetLastError());
  }
  c->path_used = 2;
  return 0;
}

template <typename V, typename L>
int dispatch_ops(fh_call* c) {
  switch (set_bits(c->op_set)) {
    case B_SUM | B_CNT: return launch_typed<V, L, B_SUM | B_CNT>(c);
    case B_SUM | B_CNT | B_PRESENT:
      return launch_typed<V, L, B_SUM | B_CNT | B_PRESENT>(c);
    case B_CNT: return launch_typed<V, L, B_CNT>(c);
    case B_MIN | B_CNT | B_PRESENT | B_NANFLAG:
      return launch_typed<V, L, B_MIN | B_CNT | B_PRESENT | B_NANFLAG>(c);
    case B_MIN | B_CNT: return launch_typed<V, L, B_MIN | B_CNT>(c);
    case B_MAX | B_CNT | B_PRESENT | B_NANFLAG:
      return launch_typed<V, L, B_MAX | B_CNT | B_PRESENT | B_NANFLAG>(c);
    case B_MAX | B_CNT: return launch_typed<V, L, B_MAX | B_CNT>(c);
    case B_SSD: return launch_typed<V, L, B_SSD>(c);
    case B_PROD | B_CNT | B_PRESENT:
      return launch_typed<V, L, B_PROD | B_CNT | B_PRESENT>(c);
    case B_IDXMIN | B_CNT | B_PRESENT:
      return launch_typed<V, L, B_IDXMIN | B_CNT | B_PRESENT>(c);
    case B_IDXMAX | B_CNT | B_PRESENT:
      return launch_typed<V, L, B_IDXMAX | B_CNT | B_PRESENT>(c);
    case B_MIN | B_CNT | B_PRESENT | B_NANFLAG | B_ARGROW:
      if constexpr (sizeof(V) == 8)
        return launch_typed<V, L, B_MIN | B_CNT | B_PRESENT | B_NANFLAG | B_ARGROW>(c);
      else
        return 11; /* 4-byte dtypes pack (enc32,row32) keys instead */
    case B_MAX | B_CNT | B_PRESENT | B_NANFLAG | B_ARGROW:
      if constexpr (sizeof(V) == 8)
        return launch_typed<V, L, B_MAX | B_CNT | B_PRESENT | B_NANFLAG | B_ARGROW>(c);
      else
        return 11;
    default: return 4;
  }
}

template <typename V>
int dispatch_label(fh_call* c) {
  switch (c->ldtype) {
    case FH_L_I64: return dispatch_ops<V, int64_t>(c);
    case FH_L_I32: return dispatch_ops<V, int32_t>(c);
    default: return 5;
  }
}


/* launch geometry for the column path: vector width per thread, column
 * blocks, and how many row chunks to split into so the chip is filled
 * (~2048 workgroups) when the column count alone is too small */
struct ColsPlan {
  int vc;
  int64_t ncolblk;
  int nchunks;
  int64_t chunk_rows;
  BinLayout lay;  /* per-chunk slab layout over ngroups*m bins */
};

template <typename V>
ColsPlan cols_plan(const fh_call* c) {
  ColsPlan p{};
  p.vc = Traits<V>::VEC;
  /* 4-byte dtypes: 8 columns per thread (two 16-B vectors) amortizes the
   * per-row LDS reads, segment compare and address arithmetic further */
  if (sizeof(V) == 4 && c->ldm % 8 == 0 && c->m >= 8 &&
      ((uintptr_t)c->values % 32) == 0)
    p.vc = 8;
  if (c->ldm % p.vc != 0 || ((uintptr_t)c->values % 16) != 0) p.vc = 1;
  p.ncolblk = (c->m + (int64_t)COLS_BLOCK * p.vc - 1) / ((int64_t)COLS_BLOCK * p.vc);
  if (p.ncolblk == 0) p.ncolblk = 1;
  int want = (int)((2048 + p.ncolblk - 1) / p.ncolblk);
  if (want < 1) want = 1;
  if (want > 64) want = 64;
  p.lay = bin_layout<V>(set_bits(c->op_set), c->ngroups * c->m, 4);
  /* cap slab at 2 GiB */
  while (want > 1 && (int64_t)want * p.lay.bytes > (int64_t)2 << 30) want--;
  if ((int64_t)want * COLS_TTILE > c->n) want = (int)((c->n + COLS_TTILE - 1) / COLS_TTILE);
  if (want < 1) want = 1;
  p.nchunks = want;
  p.chunk_rows = (c->n + p.nchunks - 1) / p.nchunks;
  return p;
}

/* init + decode shared with the atomic path, bins sized ngroups*m */
template <typename V, int OPS>
int launch_cols(fh_call* c) {
  using TR = Traits<V>;
  hipStream_t stream = (hipStream_t)c->stream;
  const int64_t nbins = c->ngroups * c->m;
  ColsPlan plan = cols_plan<V>(c);
  const bool aligned_chunks = c->chunk_offsets != nullptr && c->nchunks > 0;
  if (aligned_chunks) plan.nchunks = (int)c->nchunks;
  const bool slab_mode = !aligned_chunks && plan.nchunks > 1;
  const int skipnan = (c->flags & FH_SKIPNAN) ? 1 : 0;

  if (slab_mode) {
    if ((int64_t)plan.nchunks * plan.lay.bytes > c->scratch_bytes) return 3;
    /* identity-fill every chunk's slab sections (bins no segment touches) */
    for (int ch = 0; ch < plan.nchunks; ++ch) {
      char* s = (char*)c->scratch + (int64_t)ch * plan.lay.bytes;
      if (OPS & (B_SUM | B_SSD | B_WELFORD))
        FH_CHECK(hipMemsetAsync(s + plan.lay.sum_off, 0, nbins * 8, stream));
      if (OPS & B_WELFORD)
        FH_CHECK(hipMemsetAsync(s + plan.lay.sumx_off, 0, nbins * 8, stream));
      if (OPS & B_PROD) {
        int fb = (int)((nbins + 255) / 256);
        if (c->vdtype == FH_F32 || c->vdtype == FH_F64)
          hipLaunchKernelGGL(k_fill_f64, dim3(fb), dim3(256), 0, stream,
                             (double*)(s + plan.lay.sum_off), nbins, 1.0);
        else
          hipLaunchKernelGGL(k_fill_i64, dim3(fb), dim3(256), 0, stream,
                             (int64_t*)(s + plan.lay.sum_off), nbins, (int64_t)1);
        FH_CHECK(hipGetLastError());
      }
      if (OPS & B_CNT) FH_CHECK(hipMemsetAsync(s + plan.lay.cnt_off, 0, nbins * 4, stream));
      if (OPS & B_PRESENT)
        FH_CHECK(hipMemsetAsync(s + plan.lay.present_off, 0, nbins * 4, stream));
      if (OPS & B_MIN)
        FH_CHECK(hipMemsetAsync(s + plan.lay.minmax_off, 0xFF,
                                nbins * sizeof(typename TR::Enc), stream));
      if (OPS & B_MAX)
        FH_CHECK(hipMemsetAsync(s + plan.lay.minmax_off, 0x00,
                                nbins * sizeof(typename TR::Enc), stream));
      if (OPS & B_NANFLAG)
        FH_CHECK(hipMemsetAsync(s + plan.lay.nanflag_off, 0, nbins * 4, stream));
    }
  } else {
    if (OPS & (B_SUM | B_SSD | B_WELFORD))
      FH_CHECK(hipMemsetAsync(c->out_sum, 0, nbins * 8, stream));
    if (OPS & B_WELFORD)
      FH_CHECK(hipMemsetAsync(c->out_min, 0, nbins * 8, stream));
    if (OPS & B_PROD) {
      int fb = (int)((nbins + 255) / 256);
      if (c->vdtype == FH_F32 || c->vdtype == FH_F64)
        hipLaunchKernelGGL(k_fill_f64, dim3(fb), dim3(256), 0, stream,
                           (double*)c->out_sum, nbins, 1.0);
      else
        hipLaunchKernelGGL(k_fill_i64, dim3(fb), dim3(256), 0, stream,
                           (int64_t*)c->out_sum, nbins, (int64_t)1);
      FH_CHECK(hipGetLastError());
    }
    if (OPS & B_CNT) FH_CHECK(hipMemsetAsync(c->out_count, 0, nbins * 8, stream));
    if (OPS & B_PRESENT) FH_CHECK(hipMemsetAsync(c->out_present, 0, nbins * 4, stream));
    if (OPS & B_MIN)
      FH_CHECK(hipMemsetAsync(c->out_min, 0xFF, nbins * sizeof(typename TR::Enc), stream));
    if (OPS & B_MAX)
      FH_CHECK(hipMemsetAsync(c->out_max, 0x00, nbins * sizeof(typename TR::Enc), stream));
    if (OPS & B_NANFLAG) FH_CHECK(hipMemsetAsync(c->out_nanflag, 0, nbins * 4, stream));
  }

  dim3 grid((uint32_t)plan.ncolblk, (uint32_t)plan.nchunks);
  const int64_t* offs = aligned_chunks ? (const int64_t*)c->chunk_offsets : nullptr;
  auto launch = [&](auto kern) -> int {
    hipLaunchKernelGGL(kern, grid, dim3(COLS_BLOCK), 0, stream,
                       (const V*)c->values, (const int*)c->labels,
                       (const int*)c->perm, c->n, c->m, c->ldm, c->ngroups,
                       c->means, skipnan, plan.chunk_rows, offs,
                       (char*)c->scratch, plan.lay, c->out_sum, c->out_count,
                       c->out_present, c->out_min, c->out_max, c->out_nanflag);
    return (int)hipGetLastError();
  };
  int rc;
  const bool sk = skipnan != 0;
  if constexpr (sizeof(V) == 4) {
    if (plan.vc == 8) {
      rc = slab_mode
               ? launch(sk ? k_reduce_cols<V, OPS, 8, true, true>
                           : k_reduce_cols<V, OPS, 8, true, false>)
               : launch(sk ? k_reduce_cols<V, OPS, 8, false, true>
                           : k_reduce_cols<V, OPS, 8, false, false>);
      if (rc != 0) return rc + 1000;
      goto cols_launched;
    }
  }
  if (slab_mode)
    rc = plan.vc > 1
             ? launch(sk ? k_reduce_cols<V, OPS, Traits<V>::VEC, true, true>
                         : k_reduce_cols<V, OPS, Traits<V>::VEC, true, false>)
             : launch(sk ? k_reduce_cols<V, OPS, 1, true, true>
                         : k_reduce_cols<V, OPS, 1, true, false>);
  else
    rc = plan.vc > 1
             ? launch(sk ? k_reduce_cols<V, OPS, Traits<V>::VEC, false, true>
                         : k_reduce_cols<V, OPS, Traits<V>::VEC, false, false>)
             : launch(sk ? k_reduce_cols<V, OPS, 1, false, true>
                         : k_reduce_cols<V, OPS, 1, false, false>);
  if (rc != 0) return rc + 1000;
cols_launched:;

  if (slab_mode) {
    /* fold the chunk partials; k_combine decodes min/max (gridDim.y==1) */
    int cb = (int)((nbins + 255) / 256);
    hipLaunchKernelGGL((k_combine<V, OPS>), dim3(cb, 1), dim3(256), 0, stream,
                       (const char*)c->scratch, plan.nchunks, nbins, plan.lay,
                       c->out_sum, c->out_count, c->out_present, c->out_min,
                       c->out_max, c->out_nanflag);
    FH_CHECK(hipGetLastError());
  } else if (OPS & (B_MIN | B_MAX)) {
    int db = (int)((nbins + 255) / 256);
    hipLaunchKernelGGL((k_decode<V, OPS>), dim3(db), dim3(256), 0, stream,
                       nbins, c->out_min, c->out_max, c->out_count,
                       c->out_present);
    FH_CHECK(hipGetLastError());
  }
  c->path_used = 3;
  return 0;
}

template <typename V>
int dispatch_cols_ops(fh_call* c) {
  switch (set_bits(c->op_set)) {
    case B_SUM | B_CNT: return launch_cols<V, B_SUM | B_CNT>(c);
    case B_SUM | B_CNT | B_PRESENT:
      return launch_cols<V, B_SUM | B_CNT | B_PRESENT>(c);
    case B_CNT: return launch_cols<V, B_CNT>(c);
    case B_MIN | B_CNT | B_PRESENT | B_NANFLAG:
      return launch_cols<V, B_MIN | B_CNT | B_PRESENT | B_NANFLAG>(c);
    case B_MIN | B_CNT: return launch_cols<V, B_MIN | B_CNT>(c);
    case B_MAX | B_CNT | B_PRESENT | B_NANFLAG:
      return launch_cols<V, B_MAX | B_CNT | B_PRESENT | B_NANFLAG>(c);
    case B_MAX | B_CNT: return launch_cols<V, B_MAX | B_CNT>(c);
    case B_SSD: return launch_cols<V, B_SSD>(c);
    case B_WELFORD | B_CNT: return launch_cols<V, B_WELFORD | B_CNT>(c);
    case B_PROD | B_CNT | B_PRESENT:
      return launch_cols<V, B_PROD | B_CNT | B_PRESENT>(c);
    default: return 4;
  }
}

}  // namespace

/* pack (order-preserving 32-bit value encoding, global row) into one int64
 * key whose grouped MIN is argmin/argmax with np.argmin's first-occurrence
 * tie-break — one 12 GB/1e9-row pass replacing ~10 torch elementwise passes.
 * ismax inverts the encoding; NaN rows take the smallest key (propagate: a
 * NaN wins, as np.argmax) or all-ones (skipnan: all-NaN groups land on the
 * empty-group sentinel INT64_MAX after the sign-flip). */
template <typename V>
__global__ void k_pack_argkeys(const V* __restrict__ v, int64_t n,
                               int64_t row_offset, int ismax, int skipnan,
                               int64_t* __restrict__ out) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint64_t row = (uint64_t)(i + row_offset);
    uint64_t key;
    if constexpr (std::is_same<V, float>::value) {
      const float x = v[i];
      if (x != x) {
        key = skipnan ? ~0ULL : row;
      } else {
        uint32_t u = __float_as_uint(x);
        uint32_t enc = (u & 0x80000000u) ? ~u : (u | 0x80000000u);
        if (ismax) enc = ~enc;
        key = ((uint64_t)enc << 32) | row;
      }
    } else {
      uint32_t enc = (uint32_t)v[i] ^ 0x80000000u;
      if (ismax) enc = ~enc;
      key = ((uint64_t)enc << 32) | row;
    }
    out[i] = (int64_t)(key ^ (1ULL << 63));  /* i64 compare == u64 compare */
  }
}

extern "C" {

int64_t fh_scratch_bytes(const fh_call* c) {
  int bits = set_bits(c->op_set);
  int64_t per_block;
  switch (c->vdtype) {
    case FH_F32: per_block = bin_layout<float>(bits, c->ngroups, 4).bytes; break;
    case FH_F64: per_block = bin_layout<double>(bits, c->ngroups, 4).bytes; break;
    case FH_I64: per_block = bin_layout<int64_t>(bits, c->ngroups, 4).bytes; break;
    case FH_I32: per_block = bin_layout<int32_t>(bits, c->ngroups, 4).bytes; break;
    default: return -1;
  }
  if (c->m > 0) {
    if (c->chunk_offsets) return 0; /* group-aligned chunks write directly */
    /* column path: per-chunk slab when the row range is split */
    ColsPlan p;
    switch (c->vdtype) {
      case FH_F32: p = cols_plan<float>(c); break;
      case FH_F64: p = cols_plan<double>(c); break;
      case FH_I64: p = cols_plan<int64_t>(c); break;
      default: p = cols_plan<int32_t>(c); break;
    }
    return p.nchunks > 1 ? (int64_t)p.nchunks * p.lay.bytes : 0;
  }
  if (per_block > LDS_MAX && !(c->flags & FH_FORCE_LDS)) {
    if (c->flags & FH_FORCE_ATOMIC) return 0;
    /* partition path wants the pairs buffer + bucket directory */
    PartPlan p;
    switch (c->vdtype) {
      case FH_F32: p = part_plan<float>(c); break;
      case FH_F64: p = part_plan<double>(c); break;
      case FH_I64: p = part_plan<int64_t>(c); break;
      default: p = part_plan<int32_t>(c); break;
    }
    return p.feasible ? p.bytes : 0;
  }
  int blocks_per_cu = per_block * 2 <= LDS_MAX ? 2 : 1;
  return (int64_t)NUM_CU * blocks_per_cu * per_block;
}

int fh_grouped_reduce(fh_call* c) {
  if (!c || !c->values || !c->labels) return 6;
  if (c->labels2 && c->g0 * c->g1 != c->ngroups) return 7;
  if (c->op_set == FH_SET_SSD && !c->means) return 8;
  switch (c->vdtype) {
    case FH_F32: return dispatch_label<float>(c);
    case FH_F64: return dispatch_label<double>(c);
    case FH_I64: return dispatch_label<int64_t>(c);
    case FH_I32: return dispatch_label<int32_t>(c);
    default: return 9;
  }
}


int fh_grouped_reduce_cols(fh_call* c) {
  /* column path: values (n_t rows x m columns, column stride 1, row stride
   * ldm); labels = group codes pre-sorted ascending (int32), perm = the
   * argsort permutation (int32) such that codes_sorted[i] = codes[perm[i]].
   * Outputs are (ngroups, m) group-major. */
  if (!c || !c->values || !c->labels || !c->perm) return 6;
  if (c->op_set == FH_SET_SSD && !c->means) return 8;
  switch (c->vdtype) {
    case FH_F32: return dispatch_cols_ops<float>(c);
    case FH_F64: return dispatch_cols_ops<double>(c);
    case FH_I64: return dispatch_cols_ops<int64_t>(c);
    case FH_I32: return dispatch_cols_ops<int32_t>(c);
    default: return 9;
  }
}

int fh_pack_argkeys(const void* values, int vdtype, int64_t n,
                    int64_t row_offset, int ismax, int skipnan, void* out,
                    void* stream) {
  if (!values || !out || n < 0) return 6;
  if (n + row_offset >= (1LL << 32)) return 10; /* row must fit 32 bits */
  int64_t wb = (n + 255) / 256;
  int nblocks = (int)(wb < 8192 ? (wb > 0 ? wb : 1) : 8192);
  hipStream_t s = (hipStream_t)stream;
  if (vdtype == FH_F32)
    hipLaunchKernelGGL(k_pack_argkeys<float>, dim3(nblocks), dim3(256), 0, s,
                       (const float*)values, n, row_offset, ismax, skipnan,
                       (int64_t*)out);
  else if (vdtype == FH_I32)
    hipLaunchKernelGGL(k_pack_argkeys<int32_t>, dim3(nblocks), dim3(256), 0, s,
                       (const int32_t*)values, n, row_offset, ismax, skipnan,
                       (int64_t*)out);
  else
    return 9;
  FH_CHECK(hipGetLastError());
  return 0;
}

const char* fh_error_string(int code) {
  switch (code) {
    case 0: return "ok";
    case 2: return "FH_FORCE_LDS but bins exceed LDS capacity";
    case 3: return "scratch buffer too small (call fh_scratch_bytes)";
    case 4: return "unknown op_set";
    case 5: return "unknown label dtype";
    case 6: return "null values/labels pointer";
    case 7: return "labels2 given but g0*g1 != ngroups";
    case 8: return "FH_SET_SSD requires means";
    case 9: return "unknown value dtype";
    case 11: return "arg-pair reduction needs the bucket-partition path (8-byte dtype, n < 2^31, ngroups <= 2^24)";
    default: return code >= 1000 ? hipGetErrorString((hipError_t)(code - 1000)) : "unknown error";
  }
}

int fh_version(void) { return 1; }

}  /* extern "C" */
