/* floxhip — MI355X (gfx950) grouped-reduction engine: C ABI.
 *
 * This is the drop-in boundary for the hot path of xarray-contrib/flox
 * (reference: flox/core.py:214-394 `chunk_reduce` and the per-engine
 * aggregate modules flox/aggregate_flox.py / flox/aggregate_npg.py).
 * The Python shim flox_amd/aggregate_hip.py implements the reference's
 * engine-plugin interface (flox/aggregations.py:60-133 `generic_aggregate`:
 * one callable per reduction name, signature
 * f(group_idx, array, *, axis=-1, size=None, fill_value=None, dtype=None))
 * on top of exactly these entry points.
 *
 * One call = one fused factorize+reduce pass over `n` rows producing
 * per-group partial bins (the "intermediates" of the reference's
 * IntermediateDict, flox/types.py:28). Multi-GPU combine = an RCCL
 * all-reduce of these bins, applying the reference's combine recipes
 * (flox/aggregations.py:304-546).
 *
 * All pointers are DEVICE pointers; the caller owns every buffer and
 * synchronises via `stream` (a hipStream_t). Calls are stateless and
 * thread-safe. Errors: non-zero return, message via fh_error_string().
 */
#ifndef FLOXHIP_H
#define FLOXHIP_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* value dtypes */
enum fh_dtype { FH_F32 = 0, FH_F64 = 1, FH_I64 = 2, FH_I32 = 3 };
/* label dtypes */
enum fh_ldtype { FH_L_I64 = 0, FH_L_I32 = 1 };

/* Fused op sets: the per-group partials one pass produces.
 * (SUM accumulates float inputs in float64 — the numpy_groupies numerics
 * contract pinned by reference tests/test_properties.py:146-151.) */
enum fh_opset {
  FH_SET_SUM_COUNT = 0,         /* sum(f64/i64) + count(i64): mean, var pass 1 */
  FH_SET_SUM_COUNT_PRESENT = 1, /* + per-group "any row seen" flag: sum/nansum */
  FH_SET_COUNT = 2,             /* count only */
  FH_SET_MIN_FULL = 3,          /* min + count + present + nanflag: min */
  FH_SET_MIN_COUNT = 4,         /* min + count: nanmin */
  FH_SET_MAX_FULL = 5,          /* max + count + present + nanflag: max */
  FH_SET_MAX_COUNT = 6,         /* max + count: nanmax */
  FH_SET_SSD = 7,               /* sum of (x - mean[g])^2, f64: var pass 2 */
  FH_SET_PROD = 8,              /* product + count + present (CAS loop) */
  /* index extremum + count + present: the second pass of arg-reductions
   * (reference chunk_argreduce core.py:157-211 / argmax-argmin recipes
   * aggregations.py:582-649) and of first/last/nanfirst/nanlast. With
   * `target` set, only rows whose value equals target[group] (NaN matches
   * NaN) are candidates; without it every (valid, and non-NaN when
   * FH_SKIPNAN) row is. IDXMIN keeps the smallest row index, IDXMAX the
   * largest; out_count/out_present as usual; the index (+ row_offset, for
   * multi-GPU shards) lands in out_sum as int64, -2^63.. sentinel for
   * untouched groups handled by init (IDXMIN: INT64_MAX, IDXMAX: -1). */
  FH_SET_IDXMIN = 9,
  FH_SET_IDXMAX = 10,
  /* single-pass variance partials for the COLUMN path only: per (group,
   * column) the kernel accumulates shifted sums about the segment's first
   * value (the stability device numpy_groupies also uses,
   * aggregate_npg.py:112-126) and emits the var_chunk triple of the
   * reference (aggregations.py:348-389): ssd -> out_sum (f64),
   * sum -> out_min (f64, pointer reused), count -> out_count. */
  FH_SET_WELFORD = 11,
  /* single-call arg-reductions for 8-byte value dtypes (f64/i64) at group
   * counts beyond the LDS IDX bins, where no 32-bit value encoding can
   * pack into one int64 key: the partition pairs carry the row index in
   * their spare pad word, phase 1 is the ordinary bucketed MIN/MAX (+count
   * +present +nanflag), and a second bucket pass (k_reduce_bucket_argrow)
   * re-reads the SCATTERED pairs and takes the smallest row whose value
   * equals the group's extremum — target lookups are LDS-local after the
   * partition, unlike the two-pass form's random 80 MB gathers. Requires
   * the partition path (returns error 11 otherwise); rows must satisfy
   * row_offset + n < 2^32. Outputs: out_sum = int64 row index (INT64_MAX
   * for empty; first occurrence on ties, np.argmin semantics),
   * out_min/out_max = the decoded extremum, plus count/present/nanflag. */
  FH_SET_ARGMIN_PAIR = 12,
  FH_SET_ARGMAX_PAIR = 13,
};

/* flags */
enum fh_flags {
  FH_SKIPNAN = 1 << 0,    /* nan* reductions: NaN values contribute nothing */
  FH_FORCE_LDS = 1 << 1,  /* testing: force the LDS-binned path */
  FH_FORCE_ATOMIC = 1 << 2, /* testing: force the global-atomic path */
  FH_SORTED_LABELS = 1 << 3, /* caller guarantees labels are nondecreasing and
                              * all in [0, ngroups): fh_grouped_scan skips the
                              * radix sort (the reference's issorted fast path,
                              * aggregate_flox.py:9-23) */
  FH_NO_HOST_SYNC = 1 << 4, /* stream is being captured into a hipGraph: only
                             * paths with no D2H readback / stream sync may
                             * run (LDS, sorted-direct, atomic — NOT the
                             * bucket-partition paths, whose overflow check
                             * and counting pre-pass synchronize) */
};

typedef struct fh_call {
  int op_set;     /* fh_opset */
  int vdtype;     /* fh_dtype of `values` */
  int ldtype;     /* fh_ldtype of `labels` (and `labels2`) */
  int flags;      /* fh_flags */
  int64_t n;      /* rows */
  int64_t ngroups;
  /* 2-D groupby (reference factorize.py:102-108 _ravel_factorized):
   * labels2 != NULL means code = labels[i]*g1 + labels2[i], with either
   * label out of [0,g0)/[0,g1) meaning "not in any group" (the reference's
   * -1 sentinel). ngroups must equal g0*g1. */
  const void* values;
  const void* labels;
  const void* labels2;
  int64_t g0, g1;
  /* per-group means (f64[ngroups]) for FH_SET_SSD, else NULL */
  const double* means;

  /* outputs: each f64/i64/u32[ngroups]; only the set's members are written.
   * out_sum: f64 for float inputs (also receives the SSD result), i64 for
   *          int inputs. out_min/out_max: in VALUE dtype, +inf/-inf
   *          (INT_MAX/MIN) for empty groups. */
  void* out_sum;
  int64_t* out_count;
  uint32_t* out_present;
  void* out_min;
  void* out_max;
  uint32_t* out_nanflag;

  /* scratch: device buffer of fh_scratch_bytes() bytes (may be NULL when 0) */
  void* scratch;
  int64_t scratch_bytes;

  void* stream;   /* hipStream_t */
  int path_used;  /* out: 1 = LDS-binned, 2 = global-atomic, 3 = column */

  /* column path only (fh_grouped_reduce_cols): array is (n rows x m
   * columns), element (t,c) at values[t*ldm + c]; labels then holds the
   * int32 group codes SORTED ascending and perm the int32 argsort
   * permutation; outputs are (ngroups, m) group-major. */
  const void* perm;
  int64_t m, ldm;
  /* optional: group-aligned row chunking (int64[nchunks+1] row offsets, each
   * chunk covering disjoint groups -> chunks write final bins directly, no
   * slab); when NULL the launcher row-splits with a partials slab instead */
  const void* chunk_offsets;
  int64_t nchunks;
  /* FH_SET_IDXMIN/IDXMAX: per-group match targets (value dtype, nullable)
   * and the global row offset of this shard (multi-GPU arg-reductions) */
  const void* target;
  int64_t row_offset;
} fh_call;

/* scratch requirement for this call (0 when the global-atomic path is used) */
int64_t fh_scratch_bytes(const fh_call* c);

/* run the fused grouped reduction; returns 0 on success */
int fh_grouped_reduce(fh_call* c);

/* grouped reduce over the leading/strided axis of a multi-column array
 * (the reference's axis-subset case, core.py:272-316 + factorize.py:24-39,
 * computed without offsetting labels: one segmented pass per column) */
int fh_grouped_reduce_cols(fh_call* c);

/* pack (order-preserving 32-bit value encoding, global row index) into one
 * int64 key per row, so that a grouped MIN over the keys is argmin/argmax
 * with np.argmin's first-occurrence tie-break (the packed form of the
 * reference's argreduce, flox/aggregate_flox.py:151-187). FH_F32/FH_I32
 * values only; requires n + row_offset < 2^32. out: int64[n]. */
int fh_pack_argkeys(const void* values, int vdtype, int64_t n,
                    int64_t row_offset, int ismax, int skipnan, void* out,
                    void* stream);

const char* fh_error_string(int code);
int fh_version(void);

#ifdef __cplusplus
}
#endif
#endif /* FLOXHIP_H */
