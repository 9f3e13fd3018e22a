/* mz_gpu.h — C ABI of the MI355X-native incremental join/reduce engine.
 *
 * This is the drop-in boundary for Materialize's compute hot path
 * (SURVEY.md §8b). Each entry point states the reference interface it
 * replaces (file:line under /root/reference). A Rust host would bind these
 * over plain `extern "C"` FFI; no torch types appear
 * here — plain pointers and sizes only.
 *
 * Ownership: the caller owns all host/device memory passed in via
 * descriptors until the call returns. The library owns arrangements,
 * operators and returned out-batches; out-batches are freed with
 * mz_gpu_out_release. All calls on one context must be serialized by the
 * caller (one driver thread per GPU, mirroring one timely worker per core —
 * src/compute/src/server.rs:327-377).
 *
 * Errors: non-zero int return; mz_gpu_last_error(ctx) gives a message.
 */
#ifndef MZ_GPU_H
#define MZ_GPU_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct mz_gpu_ctx mz_gpu_ctx;
typedef struct mz_gpu_arr mz_gpu_arr;   /* an arrangement (trace/spine) */
typedef struct mz_gpu_join mz_gpu_join; /* a linear-join operator (mz_join_core state) */
typedef struct mz_gpu_red mz_gpu_red;   /* an accumulable-reduce operator */

/* ---------------------------------------------------------------- schema */

/* Arrangement schema: fixed-width keys (1..2 u64 words) and vals (bytes).
 * Blueprint: RowRowSpine columnar layout, src/row-spine/src/lib.rs:56-135. */
typedef struct {
  uint32_t key_words;  /* 1 or 2; keys compared as i64-tuple ascending */
  uint32_t val_bytes;  /* 0..64 fixed width, or MZ_GPU_VARLEN: vals are
                          variable-length byte strings compared
                          lexicographically (shorter-prefix-first), the
                          reference's byte-arena row layout
                          (row-spine/src/lib.rs:110-135)              */
} mz_gpu_schema;

#define MZ_GPU_VARLEN 0xFFFFFFFFu

/* A set of updates ((key, val), time, diff), SoA columns.
 * `on_device` = 1 when the pointers are HIP device pointers on the ctx's
 * device; 0 for host memory (the library stages them in).
 * This is both the delta-batch input format and the sealed-batch push
 * format; sealed batches must additionally be sorted by (key, val, time)
 * and consolidated (duplicate (key,val,time) diffs summed, zeros dropped) —
 * use mz_gpu_consolidate to produce that form. */
typedef struct {
  const uint64_t *keys;   /* [key_words * n] key words, key-major          */
  const uint8_t  *vals;   /* [val_bytes * n], may be NULL if val_bytes==0  */
  const uint64_t *times;  /* [n] */
  const int64_t  *diffs;  /* [n] */
  uint64_t n;
  uint64_t lower, upper;  /* batch time bounds [lower, upper)              */
  int32_t  on_device;
  /* 1 when rows are already sorted ascending by (key, val, time) in the
   * canonical order (the form mz_gpu_consolidate produces). Purely an
   * optimization hint: sorted delta streams probe large arrangements by
   * merge scan (streaming both sorted sides) instead of per-row hash
   * lookups. 0 is always safe. */
  int32_t  sorted;
  /* VARLEN schemas only: n+1 offsets into `vals` (the byte arena); row
   * i's val is vals[val_offs[i] .. val_offs[i+1]). NULL otherwise. */
  const uint32_t *val_offs;
} mz_gpu_updates;

/* ----------------------------------------------------------- closures
 * JoinClosure (src/compute-types/src/plan/join.rs:60-86) restated as a
 * filter + field-map spec evaluated inside the probe kernel. Fields are
 * copied from the probe inputs into the output (key,val) row. */
enum {
  MZ_SRC_KEY = 0,      /* the (shared) join key words                  */
  MZ_SRC_VAL_STREAM = 1, /* the delta/stream side's val bytes          */
  MZ_SRC_VAL_LOOKUP = 2, /* the arrangement side's val bytes           */
  MZ_SRC_COMPUTE = 3   /* computed field; `off` = compute id           */
};
enum { MZ_CMP_LT = 0, MZ_CMP_LE, MZ_CMP_GT, MZ_CMP_GE, MZ_CMP_EQ, MZ_CMP_NE };
enum {
  /* revenue = extendedprice_cents * (10000 - discount_bp) as i64 1e-4
   * units — Q3/Q5/Q10's `l_extendedprice * (1 - l_discount)` with TPC-H
   * fixed-scale decimals mapped to exact integers. Operand
   * offsets are given by arg0/arg1 on the out field. */
  MZ_COMPUTE_REVENUE = 0,
  /* 8 zero bytes (re-key to a constant, e.g. cross-join stages). */
  MZ_COMPUTE_CONST0 = 1,
  /* Q17's correlated-average filter, exact in integers:
   * quantity < 0.2 * sum/count  <=>  5*q*count < sum  (count > 0).
   * As a FILTER compute: arg0 = i64 quantity offset; arg1 = the offset of
   * a SUM_I64 aggregate slot's i128 value, with the COUNT slot's i64 at
   * arg1+24 (the 24-byte aggregate slot layout). NULL count (count==0 in
   * the reference's CASE, tpch_create_index.slt:1470) fails the filter. */
  MZ_COMPUTE_Q17_QTYLT = 2,
  /* i64 division as an OUT FIELD: arg0 / arg1 (both i64 at the given
   * offsets). arg1 == 0 raises MZ_ERR_DIVISION_BY_ZERO: the row is
   * diverted to the out-batch's error stream — the could_error ok/err
   * split of linear_join.rs:495-541. */
  MZ_COMPUTE_DIV_I64 = 3,
  /* i64 wrapping multiply as an OUT FIELD: arg0 * arg1 (both i64 at the
   * given offsets; Diff-style wrapping per overflowing.rs:24-31). Q6's
   * `l_extendedprice * l_discount` term. */
  MZ_COMPUTE_MUL_I64 = 4,
  /* Field-vs-field compare as a FILTER compute: passes iff
   * cmp(arg0_field, arg1_field) holds, both read at the filter's `width`
   * (4 or 8, signed LE) from (arg0_src, arg0) / (arg1_src, arg1) with
   * the filter's `cmp`. Q12's `l_commit_date < l_receipt_date` /
   * `l_ship_date < l_commit_date` predicates. */
  MZ_COMPUTE_CMP_FIELDS = 5
};

typedef struct {
  uint8_t  src;     /* MZ_SRC_* (KEY/VAL_STREAM/VAL_LOOKUP), or
                       MZ_SRC_COMPUTE with `off` = compute filter id    */
  uint16_t off;     /* byte offset into src (or compute id)            */
  uint8_t  width;   /* 4 or 8 (signed little-endian integer)           */
  uint8_t  cmp;     /* MZ_CMP_*                                        */
  int64_t  imm;     /* literal operand                                 */
  uint16_t arg0, arg1;          /* compute-filter operand offsets      */
  uint8_t  arg0_src, arg1_src;
} mz_gpu_filter;

typedef struct {
  uint8_t  src;     /* MZ_SRC_*                                        */
  uint16_t off;     /* byte offset (or compute id for MZ_SRC_COMPUTE)  */
  uint8_t  width;   /* bytes copied / produced                         */
  uint16_t arg0, arg1; /* compute operand byte offsets (both in VAL_*) */
  uint8_t  arg0_src, arg1_src;
} mz_gpu_field;

#define MZ_GPU_MAX_FILTERS 6
#define MZ_GPU_MAX_FIELDS  8

typedef struct {
  uint32_t n_filters;
  mz_gpu_filter filters[MZ_GPU_MAX_FILTERS];
  uint32_t n_key_fields;             /* output key (next stage / reduce) */
  mz_gpu_field key_fields[MZ_GPU_MAX_FIELDS];
  uint32_t n_val_fields;             /* output val                        */
  mz_gpu_field val_fields[MZ_GPU_MAX_FIELDS];
  mz_gpu_schema out;                 /* shape of the output rows          */
} mz_gpu_closure;

/* ------------------------------------------------------------- reduce
 * AccumulablePlan (src/compute-types/src/plan/reduce.rs:233) restated.
 * Accum semantics follow src/compute/src/render/reduce.rs:1611-2270:
 * SimpleNumber = wrapping i128 + non_nulls; Float = 24-frac-bit fixed-point
 * wrapping i128 + inf/nan/non_null counts; COUNT = non_nulls only. */
enum {
  MZ_AGG_COUNT = 0,
  MZ_AGG_SUM_I64,    /* also exact-decimal sums (i64 cents etc.)       */
  MZ_AGG_SUM_F64     /* fixed-point accumulation, reduce.rs:1641-1697  */
};

typedef struct {
  uint8_t  func;     /* MZ_AGG_*                                       */
  uint16_t off;      /* byte offset of the datum in the input val      */
  uint8_t  width;    /* 4 or 8                                         */
  uint8_t  is_float; /* datum is f64 (for SUM_F64)                     */
  uint8_t  nullable; /* datum has a null indicator byte at off+width   */
} mz_gpu_aggregate;

#define MZ_GPU_MAX_AGGS 4

typedef struct {
  uint32_t n_aggs;
  mz_gpu_aggregate aggs[MZ_GPU_MAX_AGGS];
  mz_gpu_schema in;   /* input (key,val) schema                        */
  mz_gpu_schema out;  /* output: key + finalized aggregate row         */
} mz_gpu_reduce_spec;

/* --------------------------------------------------------------- context */

typedef struct {
  uint64_t hbm_pool_bytes;   /* 0 = default                            */
  uint32_t device_index;
} mz_gpu_cfg;

mz_gpu_ctx *mz_gpu_init(const mz_gpu_cfg *cfg);
void        mz_gpu_fini(mz_gpu_ctx *ctx);
const char *mz_gpu_last_error(mz_gpu_ctx *ctx);
/* Synchronize the context's stream (all prior calls complete). */
int         mz_gpu_sync(mz_gpu_ctx *ctx);

/* ----------------------------------------------------------- arrangement
 * Replaces mz_arrange_core + Spine maintenance
 * (src/compute/src/extensions/arrange.rs:69-114,
 *  src/compute/src/arrangement/manager.rs:54). */
mz_gpu_arr *mz_gpu_arr_create(mz_gpu_ctx *ctx, const mz_gpu_schema *schema);
void        mz_gpu_arr_drop(mz_gpu_ctx *ctx, mz_gpu_arr *arr);
/* Push a sealed, sorted, consolidated batch (see mz_gpu_updates docs). */
int  mz_gpu_arr_push_batch(mz_gpu_ctx *ctx, mz_gpu_arr *arr,
                           const mz_gpu_updates *batch);
/* Consolidate raw updates and push the sealed batch in one call (the
 * MergeBatcher + arrange step fused — no intermediate out-batch). */
int  mz_gpu_arr_insert(mz_gpu_ctx *ctx, mz_gpu_arr *arr,
                       const mz_gpu_updates *raw);

/* Overlapped maintenance: the async form enqueues the insert's
 * consolidation+build on the arrangement's own HIP stream and returns
 * without a device sync; the sealed batch joins the spine at
 * mz_gpu_arr_flush (any probe of the arrangement flushes implicitly, as
 * does mz_gpu_sync). Independent arrangements' inserts overlap this way.
 * mz_gpu_arr_insert == insert_async + flush.
 * Lifetime: host-memory updates are staged inside the call (pageable
 * async copies complete in-call); DEVICE-pointer updates must stay
 * valid until the arrangement is flushed (explicitly or by a probe). */
int  mz_gpu_arr_insert_async(mz_gpu_ctx *ctx, mz_gpu_arr *arr,
                             const mz_gpu_updates *updates);
int  mz_gpu_arr_flush(mz_gpu_ctx *ctx, mz_gpu_arr *arr);
/* Advance the logical compaction frontier (times advance to it on merge) —
 * cf. set_logical_compaction, mz_join_core.rs:461. */
int  mz_gpu_arr_set_logical_compaction(mz_gpu_ctx *ctx, mz_gpu_arr *arr,
                                       uint64_t frontier);
/* Physical-compaction hint (mz_join_core.rs:465): a floor on batch
 * merging. This spine merges eagerly by level policy,
 * so the hint is recorded but imposes nothing — batches at or beyond
 * the frontier are already merged as the policy reaches them. */
int  mz_gpu_arr_set_physical_compaction(mz_gpu_ctx *ctx, mz_gpu_arr *arr,
                                        uint64_t frontier);
/* Perform up to `fuel` rows of spine merge work — cf. manager.rs:54. */
int  mz_gpu_arr_maintain(mz_gpu_ctx *ctx, mz_gpu_arr *arr, uint64_t fuel);
/* Introspection (arrangement-size logging, extensions/arrange.rs:249). */
int  mz_gpu_arr_stats(mz_gpu_ctx *ctx, mz_gpu_arr *arr, uint64_t *n_batches,
                      uint64_t *n_updates, uint64_t *hbm_bytes);

/* ------------------------------------------------------------- out batch */
typedef struct {
  uint64_t *keys;  uint8_t *vals;  uint64_t *times;  int64_t *diffs;
  uint64_t n;
  int32_t on_device;      /* 1: device pointers (default)              */
  mz_gpu_schema schema;
  /* Error-row stream: the ok/err split of the reference's join closure
   * (JoinClosure::could_error, linear_join.rs:495-541). A closure field
   * whose evaluation errors (e.g. MZ_COMPUTE_DIV_I64 by zero) diverts
   * the would-be output row here as (error code, time, diff),
   * consolidated like any update stream. err_* are device arrays owned
   * by the out-batch (NULL/0 when the closure cannot error). */
  uint64_t err_n;
  uint64_t *err_codes;    /* MZ_ERR_* per row */
  uint64_t *err_times;
  int64_t  *err_diffs;
  /* VARLEN schemas only (schema.val_bytes == MZ_GPU_VARLEN): n+1
   * offsets into `vals`; val_arena_bytes = val_offs[n]. */
  uint32_t *val_offs;
  uint64_t val_arena_bytes;
} mz_gpu_out;

enum { MZ_ERR_DIVISION_BY_ZERO = 1 };

/* VARLEN out-batches: copy the [n+1] val offsets to the host; `vals`
 * passed to mz_gpu_out_to_host must then be sized val_arena_bytes. */
int mz_gpu_out_voffs_to_host(mz_gpu_ctx *ctx, const mz_gpu_out *out,
                             uint32_t *offs);

/* Copy an out-batch's error rows to caller host buffers (sized err_n). */
int mz_gpu_out_err_to_host(mz_gpu_ctx *ctx, const mz_gpu_out *out,
                           uint64_t *codes, uint64_t *times,
                           int64_t *diffs);

/* Flush like mz_gpu_arr_flush, and additionally hand back the pending
 * insert's CONSOLIDATED flat rows as a sorted out-batch (*out = NULL when
 * no insert was pending or it was empty). This is the arrangement's
 * update stream: the same sealed rows the reference's mz_arrange_core
 * publishes to downstream operators (extensions/arrange.rs:69-114) —
 * delta-path probes consume it with `sorted = 1`. Caller releases. */
int mz_gpu_arr_flush_take(mz_gpu_ctx *ctx, mz_gpu_arr *arr,
                          mz_gpu_out **out);

void mz_gpu_out_release(mz_gpu_ctx *ctx, mz_gpu_out *out);
/* Copy an out-batch's columns to caller host buffers (sized n). */
int  mz_gpu_out_to_host(mz_gpu_ctx *ctx, const mz_gpu_out *out,
                        uint64_t *keys, uint8_t *vals, uint64_t *times,
                        int64_t *diffs);

/* Sort by (key,val,time), consolidate diffs, drop zeros — DD
 * consolidate_updates as used at mz_join_core.rs:604. */
int  mz_gpu_consolidate(mz_gpu_ctx *ctx, const mz_gpu_schema *schema,
                        const mz_gpu_updates *in, mz_gpu_out **out);

/* ------------------------------------------------------------ linear join
 * Replaces mz_join_core (mz_join_core.rs:57-496). The operator tracks the
 * acknowledged frontier of each side implicitly: a push probes the opposing
 * arrangement AS OF the call. Push concurrent batches side-1-first to
 * reproduce the reference's drain order (exactly-once). */
mz_gpu_join *mz_gpu_join_create(mz_gpu_ctx *ctx, mz_gpu_arr *arr1,
                                mz_gpu_arr *arr2, const mz_gpu_closure *cl);
void mz_gpu_join_drop(mz_gpu_ctx *ctx, mz_gpu_join *op);
/* Join `delta` (side = 1 or 2, already consolidated) against the opposing
 * arrangement; output consolidated. NOTE: push the delta to its own
 * arrangement via mz_gpu_arr_push_batch BEFORE or AFTER this call per the
 * reference discipline: arr_push(arr1,b1); join_push(1,b1);
 * arr_push(arr2,b2); join_push(2,b2). */
int  mz_gpu_join_push(mz_gpu_ctx *ctx, mz_gpu_join *op, int side,
                      const mz_gpu_updates *delta, mz_gpu_out **out);

/* ------------------------------------------------------------- half join
 * Replaces half_join2 (delta_join.rs:500,544): probe `delta`'s updates
 * (whose `times` are the promoted data-times) against `lookup`; a trace
 * update at t' matches a stream update at t iff t' <= t (le=1; source
 * relation precedes lookup relation) or t' < t (le=0) —
 * delta_join.rs:362,372. Output time = t. Output consolidated.
 * `stream_val_bytes` = the delta updates' val stride (the stream row was
 * re-keyed/thinned by the previous stage and need not match `lookup`'s). */
int  mz_gpu_halfjoin(mz_gpu_ctx *ctx, mz_gpu_arr *lookup,
                     const mz_gpu_updates *delta, uint32_t stream_val_bytes,
                     int le, const mz_gpu_closure *cl, mz_gpu_out **out);

/* Fused two-stage delta path (one path's two lookup stages,
 * delta_join.rs:338-472, in a single kernel): delta -> lookup1 (le1,
 * closure1 produces the intermediate key/val in registers) -> lookup2
 * (le2, closure2 produces the output). Equivalent to two halfjoin calls
 * with the intermediate stream never materialized; output is RAW
 * (unconsolidated — the consumer consolidates) with the err stream
 * attached. Constraints: fixed-width lookups; closure1's output key
 * must be lookup2's key schema, its output <= 2 key words / 48 val
 * bytes. */
int  mz_gpu_halfjoin2(mz_gpu_ctx *ctx, mz_gpu_arr *lookup1, int le1,
                      const mz_gpu_closure *cl1, mz_gpu_arr *lookup2,
                      int le2, const mz_gpu_closure *cl2,
                      const mz_gpu_updates *delta,
                      uint32_t stream_val_bytes, mz_gpu_out **out);

/* --------------------------------------------------------------- reduce
 * Replaces build_accumulable + mz_reduce_abelian (reduce.rs:1357-1581,
 * extensions/reduce.rs:131). The operator owns the resident accumulator
 * table AND the output arrangement; each push returns output corrections
 * (new minus old finalized rows, diffs ±1) per changed key. Input updates
 * are (key, input-val) rows; the datum→accumulator move (explode_one,
 * reduce.rs:1409-1431) happens inside. Multi-timestamp batches are
 * processed in time order. */
mz_gpu_red *mz_gpu_reduce_create(mz_gpu_ctx *ctx,
                                 const mz_gpu_reduce_spec *spec);
void mz_gpu_reduce_drop(mz_gpu_ctx *ctx, mz_gpu_red *op);
int  mz_gpu_reduce_push(mz_gpu_ctx *ctx, mz_gpu_red *op,
                        const mz_gpu_updates *delta, mz_gpu_out **out);

/* Diagnostics: when the environment sets MZ_GPU_PROF=1, the engine
 * records per-phase HIP event pairs; this prints and resets the sums
 * ("MZPROF <phase> <ms> <count>" lines on stdout). No-op otherwise. */
void mz_gpu_prof_dump(mz_gpu_ctx *ctx);

/* ----------------------------------------------------------- threshold
 * Replaces build_threshold_basic / threshold_local
 * (src/compute/src/render/threshold.rs:34-51,75-97): a reduce over the
 * row-keyed arrangement that keeps each record whose accumulated count is
 * positive, with that count as the output multiplicity
 * (count.is_positive() filter, threshold.rs:42). The operator owns the
 * resident per-(key,val) net-count table; each push returns corrections
 * with diff = pos(new_count) - pos(old_count) per changed record
 * (reduce_abelian contract, src/compute/src/extensions/reduce.rs:131).
 * Output schema equals the input schema. */
typedef struct mz_gpu_thr mz_gpu_thr;
mz_gpu_thr *mz_gpu_threshold_create(mz_gpu_ctx *ctx,
                                    const mz_gpu_schema *schema);
int  mz_gpu_threshold_push(mz_gpu_ctx *ctx, mz_gpu_thr *op,
                           const mz_gpu_updates *delta, mz_gpu_out **out);
void mz_gpu_threshold_drop(mz_gpu_ctx *ctx, mz_gpu_thr *op);

/* ----------------------------------------------------------------- topk
 * Replaces render_topk's Basic plan path — build_topk /
 * build_topk_negated_stage (src/compute/src/render/top_k.rs:322-418,
 * 614-770): per group-key, order records by the order columns
 * (ColumnOrder asc/desc; compare_columns at :733-739, ties broken in the
 * engine's canonical val order, standing in for the reference's Row-order
 * tie-break :738), then keep the multiplicity window [offset,
 * offset+limit) of the running prefix (:743-766). The operator owns the
 * resident group-contents state; each push returns corrections (new
 * minus old kept rows per changed group, reduce_abelian contract).
 * Restrictions vs the reference: literal limits only (no
 * per-key limit expressions, :659-688); order columns are non-null signed
 * little-endian integers of width 4 or 8; the monotonic plan variants
 * (MonotonicTop1/TopK, :157-288) are streaming-input optimizations whose
 * output equals Basic's and are not separate entry points; the bucketed
 * stage hierarchy (:380-398) is a work-thinning policy that leaves the
 * final modulus-1 stage's output unchanged, so groups are evaluated
 * directly. Negative input multiplicities return an error
 * ("Negative multiplicities in TopK", :494). */
typedef struct {
  uint16_t off;    /* byte offset of the order datum in the val bytes */
  uint8_t  width;  /* 4 or 8 (signed little-endian integer)           */
  uint8_t  desc;   /* 1 = descending                                  */
} mz_gpu_order_col;

#define MZ_GPU_MAX_ORDER 4

typedef struct {
  mz_gpu_schema in;  /* group key words + record val bytes            */
  uint64_t offset;   /* rows to skip per group (TopKPlan::offset)     */
  int64_t  limit;    /* rows to keep after offset; < 0 = no limit     */
  uint32_t n_order;
  mz_gpu_order_col order[MZ_GPU_MAX_ORDER];
} mz_gpu_topk_spec;

typedef struct mz_gpu_topk mz_gpu_topk;
mz_gpu_topk *mz_gpu_topk_create(mz_gpu_ctx *ctx,
                                const mz_gpu_topk_spec *spec);
int  mz_gpu_topk_push(mz_gpu_ctx *ctx, mz_gpu_topk *op,
                      const mz_gpu_updates *delta, mz_gpu_out **out);
void mz_gpu_topk_drop(mz_gpu_ctx *ctx, mz_gpu_topk *op);

/* ------------------------------------------------------------ exchange
 * Replaces the Exchange pact routing (linear_join.rs:390,
 * extensions/arrange.rs:134): shard = splitmix64(key words) % nshards
 * (hash substitution per DESIGN.md §2.2). Writes each update into its
 * shard's contiguous region of the caller-provided output columns (device
 * or host, matching `in->on_device`) and fills counts[nshards]. The
 * collective itself is the caller's (RCCL all-to-all-v over xGMI). */
int  mz_gpu_partition(mz_gpu_ctx *ctx, const mz_gpu_schema *schema,
                      const mz_gpu_updates *in, uint32_t nshards,
                      uint64_t *out_keys, uint8_t *out_vals,
                      uint64_t *out_times, int64_t *out_diffs,
                      uint64_t *counts);

/* -------------------------------------------------------------- flat map
 * FlatMap / key-preparation analog (src/compute/src/render/flat_map.rs;
 * DeltaJoinKeyPreparation, delta_join.rs:444-464): apply a closure to a
 * stream without a lookup (VAL_STREAM = the input val). */
int  mz_gpu_map(mz_gpu_ctx *ctx, const mz_gpu_schema *in,
                const mz_gpu_updates *updates, const mz_gpu_closure *cl,
                mz_gpu_out **out);

/* --------------------------------------------------- hierarchical reduce
 * Replaces build_bucketed/build_monotonic + ReductionMonoid
 * (src/compute/src/render/reduce.rs:850-1224, :2273): MIN/MAX maintained
 * through a val-hash bucket reduction tree (plan/reduce.rs:319-326);
 * retracting the current extremum recomputes only the affected buckets.
 * Input vals are single i64 datums; pushes are single-timestamp.
 * `buckets` e.g. {4096, 256, 16, 1} (last level = per key). */
typedef struct mz_gpu_minmax mz_gpu_minmax;
mz_gpu_minmax *mz_gpu_minmax_create(mz_gpu_ctx *ctx,
                                    const mz_gpu_schema *in, int is_max,
                                    const uint32_t *buckets,
                                    uint32_t n_levels);
int  mz_gpu_minmax_push(mz_gpu_ctx *ctx, mz_gpu_minmax *op,
                        const mz_gpu_updates *delta, mz_gpu_out **out);
void mz_gpu_minmax_drop(mz_gpu_ctx *ctx, mz_gpu_minmax *op);

/* ------------------------------------------------------------------ peek
 * Replaces the peek path (handle_peek/process_peeks,
 * src/compute/src/compute_state.rs:763,1155): read, for each requested
 * key, the arrangement's (val, summed diff) pairs as of `time` (updates at
 * t' <= time accumulate; zero-sum vals are dropped). `keys` are n_keys
 * host key rows. Out rows: (key, val, time, diff). */
int  mz_gpu_peek(mz_gpu_ctx *ctx, mz_gpu_arr *arr, const uint64_t *keys,
                 uint64_t n_keys, uint64_t time, mz_gpu_out **out);

/* The routing hash itself (host helper; device code uses the same). */
uint64_t mz_gpu_route_hash(const uint64_t *key_words, uint32_t n_words);

#ifdef __cplusplus
}
#endif
#endif /* MZ_GPU_H */
