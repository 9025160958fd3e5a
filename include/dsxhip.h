/* dsxhip — C ABI of the MI355X-native physical execution layer for dask-sql's
 * hot path.
 *
 * Each entry point cites the reference interface it replaces
 * (paths relative to /root/reference). The reference's own execution layer is
 * pandas/Dask called from dask_sql/physical/; this library is the
 * HIP/CDNA4 (gfx950) replacement for exactly those calls. Host bindings:
 * ctypes (dask_sql_amd/runtime.py); a cgo/JNI/N-API-style stub is shown in
 * INTEGRATION.md.
 *
 * Conventions:
 *  - plain pointers + lengths; no torch/Arrow types in signatures. Device
 *    pointers are HIP device memory on the context's device.
 *  - validity masks are uint8[n] (1 = valid); NULL pointer = all valid.
 *    (Arrow bitmaps are converted at the Python boundary.)
 *  - outputs with data-dependent size are library-allocated (hipMalloc);
 *    release with dsx_free. Fixed-size outputs are caller-allocated.
 *  - return: 0 = OK, <0 = error; dsx_last_error() has the message.
 *  - thread-safety: one DsxCtx per thread/GPU; calls on one ctx serialize on
 *    its HIP stream. Multi-GPU collectives are the host's job (RCCL via
 *    torch.distributed, one process per GPU) — this library is single-device.
 */
#ifndef DSXHIP_H
#define DSXHIP_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct DsxCtx DsxCtx;

/* dtype tags (subset per SURVEY.md §2 "Type mappings" row) */
enum DsxType {
  DSX_I64 = 0,
  DSX_F64 = 1,
  DSX_I32 = 2,    /* also DATE32 (days since epoch) */
  DSX_F32 = 3,
  DSX_I8  = 4,    /* also dictionary codes */
  DSX_BOOL8 = 5,
};

/* expression VM opcodes — the device-compiled form of the reference's Rex
 * operator subset (dask_sql/physical/rex/core/call.py:1047-1156: comparisons,
 * AND/OR/NOT, +,-,*,/, IS [NOT] NULL, CASE-as-select, CAST). Programs are
 * typed postfix; built by dask_sql_amd/physical/rex compiler. */
enum DsxOp {
  DSX_OP_COL = 1,       /* arg0 = column index; pushes typed value+validity  */
  DSX_OP_LIT_F64 = 2,   /* imm = f64 bits                                    */
  DSX_OP_LIT_I64 = 3,   /* imm = i64                                         */
  DSX_OP_LIT_NULL = 4,
  DSX_OP_ADD_F64 = 10, DSX_OP_SUB_F64 = 11, DSX_OP_MUL_F64 = 12, DSX_OP_DIV_F64 = 13,
  DSX_OP_ADD_I64 = 14, DSX_OP_SUB_I64 = 15, DSX_OP_MUL_I64 = 16,
  DSX_OP_DIV_I64 = 17, DSX_OP_MOD_I64 = 18,
  DSX_OP_LT_F64 = 20, DSX_OP_LE_F64 = 21, DSX_OP_GT_F64 = 22, DSX_OP_GE_F64 = 23,
  DSX_OP_EQ_F64 = 24, DSX_OP_NE_F64 = 25,
  DSX_OP_LT_I64 = 30, DSX_OP_LE_I64 = 31, DSX_OP_GT_I64 = 32, DSX_OP_GE_I64 = 33,
  DSX_OP_EQ_I64 = 34, DSX_OP_NE_I64 = 35,
  DSX_OP_AND = 40, DSX_OP_OR = 41, DSX_OP_NOT = 42,   /* SQL 3-valued logic   */
  DSX_OP_IS_NULL = 43, DSX_OP_IS_NOT_NULL = 44,
  DSX_OP_I64_TO_F64 = 50, DSX_OP_F64_TO_I64 = 51,  /* CAST (trunc,
                                                      mappings.py:346-353) */
  DSX_OP_BITS_F64 = 52,  /* reinterpret i64 bits as f64 (float key unpack) */
  DSX_OP_SELECT = 60,   /* (cond, a, b) -> cond ? a : b — CASE WHEN           */
  DSX_OP_NEG_F64 = 61, DSX_OP_NEG_I64 = 62, DSX_OP_SQRT_F64 = 63,
  /* scalar math + date extraction (rex/core/call.py scalar operations) */
  DSX_OP_ABS_I64 = 64, DSX_OP_ABS_F64 = 65,
  DSX_OP_FLOOR_F64 = 66, DSX_OP_CEIL_F64 = 67,
  DSX_OP_RINT_F64 = 68,  /* ties-to-even like numpy round */
  DSX_OP_EXP_F64 = 69, DSX_OP_LN_F64 = 70, DSX_OP_POW_F64 = 71,
  DSX_OP_YEAR = 72, DSX_OP_MONTH = 73, DSX_OP_DAY = 74, /* date32 day-int */
  DSX_OP_FLOORMOD_I64 = 75, /* Python/pandas floor-mod: MOD(-5,3) = 1
                               (reference evaluates operator.mod on pandas,
                               rex/core/call.py:1047-1156) */
  /* trigonometry (rex/core/call.py TrigonometricOperations — da.sin etc.);
     unary ops take/return f64, ATAN2 is binary */
  DSX_OP_SIN_F64 = 76, DSX_OP_COS_F64 = 77, DSX_OP_TAN_F64 = 78,
  DSX_OP_ASIN_F64 = 79, DSX_OP_ACOS_F64 = 80, DSX_OP_ATAN_F64 = 81,
  DSX_OP_ATAN2_F64 = 82,
};

typedef struct DsxInstr {
  int32_t op;       /* DsxOp */
  int32_t arg0;     /* column index for OP_COL */
  int64_t imm;      /* literal bits */
} DsxInstr;

#define DSX_MAX_PROG 120
#define DSX_MAX_COLS 16
#define DSX_MAX_AGGS 16
#define DSX_MAX_KEYS 4

typedef struct DsxColumn {
  void* data;               /* device pointer */
  const uint8_t* validity;  /* device pointer or NULL (all valid) */
  int64_t len;
  int32_t dtype;            /* DsxType */
} DsxColumn;

/* ---- context / memory -------------------------------------------------- */

/* replaces: process/worker setup the reference delegates to dask.distributed
 * (SURVEY.md §5 "Distributed communication backend"). One ctx per GPU. */
int dsx_ctx_create(int device_id, DsxCtx** out);
void dsx_ctx_destroy(DsxCtx* ctx);
const char* dsx_last_error(void);
int dsx_synchronize(DsxCtx* ctx);

int dsx_malloc(DsxCtx* ctx, int64_t bytes, void** out);
int dsx_free(DsxCtx* ctx, void* ptr);
/* replaces the host→worker data movement of dask's task shuffle for table
 * registration (dask_sql/context.py:168 create_table / persist):
 * pinned-host→HBM hipMemcpyAsync. */
int dsx_upload(DsxCtx* ctx, const void* host, int64_t bytes, void** out_dev);
/* host→HBM through a persistent pinned staging arena (chunked, overlapped
 * memcpy + hipMemcpyAsync) — the parquet-ingest fast path (reference reads
 * via dask IO, physical/utils/filter.py:17; here: pyarrow column buffers →
 * pinned → HBM with no pandas round-trip). */
int dsx_upload_pinned(DsxCtx* ctx, const void* host_data, int64_t nbytes,
                      void** out_device_ptr);

int dsx_download(DsxCtx* ctx, const void* dev, void* host, int64_t bytes);
int dsx_memset(DsxCtx* ctx, void* dev, int value, int64_t bytes);
/* device-to-device copy (UNION ALL concatenation — the reference's
 * dd.concat of the union branches) */
int dsx_copy(DsxCtx* ctx, void* dst, const void* src, int64_t bytes);

/* per-kernel HIP-event timing (for bench.py roofline accounting) */
int dsx_prof_enable(DsxCtx* ctx, int enable);
/* fills (name, total_ms, launches) for up to cap kernels; returns count */
int dsx_prof_get(DsxCtx* ctx, char names[][32], double* total_ms,
                 int64_t* launches, int cap);
int dsx_prof_reset(DsxCtx* ctx);

/* ---- scan / expression / filter ---------------------------------------- */

/* replaces Projection `df.assign(RexConverter result)`
 * (dask_sql/physical/rel/logical/project.py:56-65 and rex/core/call.py ops):
 * out = program(cols) per row. out_dtype in {DSX_I64, DSX_F64, DSX_BOOL8}.
 * out_data caller-allocated (n * sizeof), out_validity caller-allocated u8[n]
 * or NULL to discard. */
int dsx_eval(DsxCtx* ctx, const DsxInstr* prog, int prog_len,
             const DsxColumn* cols, int ncols, int64_t n,
             void* out_data, uint8_t* out_validity, int32_t out_dtype);

/* replaces Filter `cond.fillna(False); df[cond]`
 * (dask_sql/physical/rel/logical/filter.py:20-45): evaluates the predicate
 * program, NULL→False, and emits the ORDER-PRESERVING selection vector of
 * matching row ids (library-allocated u32; free with dsx_free). */
int dsx_filter(DsxCtx* ctx, const DsxInstr* prog, int prog_len,
               const DsxColumn* cols, int ncols, int64_t n,
               uint32_t** out_sel, int64_t* out_count);

/* fused filter + materialization: evaluate the predicate and write the
 * selected rows of `mats` directly, order-preserving — `df[cond]`
 * (filter.py:40) in one pass, with no selection vector and no per-column
 * gathers. out_datas/out_valids are caller arrays of nmats slots, filled
 * with pool allocations (free with dsx_free). */
int dsx_filter_cols(DsxCtx* ctx, const DsxInstr* prog, int prog_len,
                    const DsxColumn* cols, int ncols, int64_t n,
                    const DsxColumn* mats, int nmats, void** out_datas,
                    uint8_t** out_valids, int64_t* out_count);

/* boolean-mask take / merge materialization: out[i] = col[sel[i]].
 * out caller-allocated. Gathers validity too when both non-NULL. */
int dsx_gather(DsxCtx* ctx, const DsxColumn* col, const uint32_t* sel,
               int64_t n_sel, void* out_data, uint8_t* out_validity);

/* inverse of dsx_gather: out[sel[i]] = col[i] — places join-back / window
 * columns into original row order (reference window.py:212-428 assigns the
 * computed window column back onto the frame's index). out caller-allocated
 * and pre-initialized (untouched rows keep their init). */
int dsx_scatter_rows(DsxCtx* ctx, const DsxColumn* col, const uint32_t* sel,
                     int64_t n_sel, int64_t n_out, void* out_data,
                     uint8_t* out_validity);

/* min/max of an i64/i32/i8/date32 column ignoring NULLs (key-range probe for
 * packing; also MIN/MAX aggregate support). */
int dsx_minmax_i64(DsxCtx* ctx, const DsxColumn* col, int64_t* out_min,
                   int64_t* out_max, int64_t* out_nonnull);

/* pack up to 4 key columns into one u64 code column:
 * code = Σ_k (col_k - min_k + nullable_k) * stride_k, NULL → 0 slot.
 * Implements composite GROUP BY / join keys; NULL gets its own code, which
 * is what gives groupby(dropna=False) (aggregate.py:575-577) for free.
 * out caller-allocated u64[n]. */
typedef struct DsxKeySpec {
  int32_t col;        /* index into cols */
  int64_t min;        /* from dsx_minmax */
  int64_t range;      /* max-min+1 (+1 more reserved internally if nullable) */
  int32_t nullable;   /* 0/1 */
  int32_t mode;       /* 0 = radix pack; 1 = f64 bit-pattern (must be the
                         ONLY key): code = canonical_bits(x)+1, NaN and NULL
                         share code 0 — pandas groups NaN keys together under
                         dropna=False (aggregate.py:575-577) */
} DsxKeySpec;
int dsx_keypack(DsxCtx* ctx, const DsxColumn* cols, int ncols,
                const DsxKeySpec* keys, int nkeys, int64_t n,
                uint64_t* out_codes);

/* ---- hash join ---------------------------------------------------------- */

/* replaces the per-partition pandas hash join inside
 * `dd.merge(on=..., how=...)` (dask_sql/physical/rel/logical/join.py:241-246).
 * Build: open-addressing multimap over u64 key codes (from dsx_keypack, or
 * raw non-negative i64 keys). NULL keys (code with validity 0) are NOT
 * inserted — the NULL-key drop of join.py:202-213 for the build side.
 * Table is library-allocated; free with dsx_hash_table_free. */
typedef struct DsxHashTable DsxHashTable;
/* code_max: max possible key code (0 = unknown). codes below 2^32-1 pack
 * (code,rowid) into one 8-byte slot — one random read per probe. */
int dsx_hash_build(DsxCtx* ctx, const uint64_t* codes, const uint8_t* validity,
                   int64_t n, uint64_t code_max, DsxHashTable** out);
void dsx_hash_table_free(DsxHashTable* t);

enum DsxJoinType {  /* reference join.py:41-48 JOIN_TYPE_MAPPING */
  DSX_JOIN_INNER = 0,
  DSX_JOIN_LEFT = 1,       /* left outer: unmatched probe → rhs NULL */
  DSX_JOIN_LEFTSEMI = 2,
  DSX_JOIN_LEFTANTI = 3,
};
#define DSX_NULL_IDX 0xFFFFFFFFu

/* Probe: emits (probe_rowid, build_rowid) pairs, library-allocated.
 * Unordered (normalize by sort for comparisons); FULL OUTER is composed by
 * the host from LEFT + unmatched-build sweep (dsx_hash_unmatched). */
/* mark_matched: record matched build slots (needed only when
 * dsx_hash_unmatched will run, i.e. FULL OUTER). */
int dsx_hash_probe(DsxCtx* ctx, DsxHashTable* t, const uint64_t* codes,
                   const uint8_t* validity, int64_t n, int join_type,
                   int mark_matched,
                   uint32_t** out_probe_idx, uint32_t** out_build_idx,
                   int64_t* out_count);
/* build rows never matched by any probe since build (for FULL OUTER,
 * join.py JOIN_TYPE_MAPPING "FULL" → outer). */
/* fused probe-emit + materialization (INNER/LEFT/SEMI/ANTI, no residual):
 * the emit pass writes the join's output columns directly — one pass, no
 * (probe,build) pair vectors and no per-column gathers (the fusion of
 * join.py:241-246 dd.merge's column copy). out_datas/out_valids are caller
 * arrays of n_pcols+n_bcols slots, filled with pool allocations (free with
 * dsx_free); out_valids[i] NULL when the source has no validity and no
 * LEFT NULL-fill applies. */
int dsx_hash_probe_cols(DsxCtx* ctx, DsxHashTable* t, const uint64_t* codes,
                        const uint8_t* validity, int64_t n, int join_type,
                        const DsxColumn* pcols, int n_pcols,
                        const DsxColumn* bcols, int n_bcols,
                        int force_build_validity, void** out_datas,
                        uint8_t** out_valids, int64_t* out_count);

int dsx_hash_unmatched(DsxCtx* ctx, DsxHashTable* t, uint32_t** out_build_idx,
                       int64_t* out_count);

/* Radix-partitioned equijoin (INNER/LEFT/LEFTANTI): both sides hash-
 * partitioned into bucket-major records, per-bucket LDS build+probe, fused
 * column emit. Replaces dd.merge's per-partition hash join
 * (dask_sql/physical/rel/logical/join.py:241-246) at sizes where the flat
 * probe table spills the XCD L2. keys_b/keys_p: per-key specs (col index
 * into the side's column array; min/range/nullable MUST match pairwise).
 * bpred: build-side row predicate (e.g. key IS NOT NULL). out_side[i]:
 * 0 = probe column, 1 = build column; out_col[i]: column index in that
 * side's array; out_need_valid[i]: allocate a validity mask for output i.
 * Returns -6 on bucket overflow (key skew) — caller falls back to the
 * flat-table join. */
int dsx_radix_join(DsxCtx* ctx, const DsxColumn* build_cols, int n_build_cols,
                   int64_t n_build, const DsxKeySpec* keys_b,
                   const DsxKeySpec* keys_p, int nkeys,
                   const DsxInstr* bpred, int bpred_len,
                   const DsxColumn* probe_cols, int n_probe_cols,
                   int64_t n_probe, int join_type, const int32_t* out_side,
                   const int32_t* out_col, const int32_t* out_need_valid,
                   int n_out, void** out_datas, uint8_t** out_valids,
                   int64_t* out_count);

/* Device general sort for ORDER BY without LIMIT (reference
 * physical/utils/sort.py:9-60): order-preserving packed codes (DsxKeySpec
 * mode bit1 = DESC, bit2 = NULLS LAST; caller passes keys REVERSED so the
 * first ORDER BY key takes the highest stride) → range partition →
 * per-bucket LDS bitonic (stable via rowid tiebreak). Returns the sorted
 * row permutation; -6 on skew (caller sorts on host). */
int dsx_sort_perm(DsxCtx* ctx, const DsxColumn* cols, int ncols,
                  const DsxKeySpec* keys, int nkeys, int64_t n,
                  uint32_t** out_perm);

/* Device ordered window frames (reference rel/logical/window.py:212-428):
 * caller provides the sort permutation over (partition, order) keys and the
 * partition / partition+order codes gathered into sorted order; computes
 * ROW_NUMBER/RANK/DENSE_RANK/LAG/LEAD/FIRST_VALUE and the running
 * SUM/COUNT/MIN/MAX (RANGE UNBOUNDED..CURRENT with peer broadcast),
 * scattered back to original row order. func = DsxWinFunc (0..9). */
int dsx_window_ordered(DsxCtx* ctx, const uint32_t* perm, int64_t n,
                       const uint64_t* pcode_sorted,
                       const uint64_t* fcode_sorted, int func,
                       const DsxColumn* value_or_null, int64_t offset,
                       int64_t default_bits, int has_default, int out_dtype,
                       void** out_data, uint8_t** out_valid, int want_valid);

/* ---- hash groupby-aggregate --------------------------------------------- */

enum DsxAggOp {  /* reference AGGREGATION_MAPPING aggregate.py:117-231 subset:
                    sum (custom_sum min_count=1, :486-493), count, min, max;
                    avg finalized on host as sum/count. */
  DSX_AGG_SUM_F64 = 0,
  DSX_AGG_SUM_I64 = 1,
  DSX_AGG_COUNT = 2,     /* counts rows where input program is non-NULL */
  DSX_AGG_MIN_F64 = 3,
  DSX_AGG_MAX_F64 = 4,
  DSX_AGG_MIN_I64 = 5,
  DSX_AGG_MAX_I64 = 6,
};

typedef struct DsxAggSpec {
  int32_t op;                    /* DsxAggOp */
  int32_t prog_len;
  DsxInstr prog[DSX_MAX_PROG];   /* input expression, fused into the kernel */
} DsxAggSpec;

/* Toggle the per-table groupby histogram cache; the Python layer turns it
 * off for externally-backed (e.g. torch/RCCL staging) key columns. */
int dsx_gb_hist_cache_enable(DsxCtx* ctx, int enable);

/* replaces `df.groupby(by, dropna=False).agg(...)`
 * (dask_sql/physical/rel/logical/aggregate.py:575-581) with the WHERE
 * predicate fused in (filter.py:20-45 fused into the same scan — SURVEY §3
 * call stack (2)+(4)) and the key pack fused in (one fused scan kernel).
 *
 * keys: nkeys DsxKeySpec (ranges from dsx_minmax); the packed key space
 *       (Π(range+nullable)) must be ≤ 2^62. nkeys 0 = full-table aggregate.
 *       key_space ≤ LDS budget → per-CU LDS direct-indexed accumulation;
 *       else global CAS-claim table (SURVEY §7 step 4 two-level design).
 * pred: optional predicate program (NULL→False), pred_len 0 = no predicate.
 * Outputs (library-allocated, compacted, one row per non-empty group):
 *   out_codes u64[G], out_vals = one buffer laid out [naggs][G] (f64/i64 per
 *   agg op), out_counts = one buffer [naggs][G] of non-NULL input counts
 *   (for SUM NULL semantics and COUNT), G = *out_groups.
 *   For DSX_AGG_SUM_*: value is the sum over non-NULL inputs; nonnull count 0
 *   ⇒ SQL NULL (custom_sum min_count=1) — host finalizes.
 *   For MIN/MAX: same. For COUNT: value array unused, count is the result. */
int dsx_hash_groupby(DsxCtx* ctx,
                     const DsxColumn* cols, int ncols, int64_t n,
                     const DsxKeySpec* keys, int nkeys,
                     const DsxInstr* pred, int pred_len,
                     const DsxAggSpec* aggs, int naggs,
                     uint64_t** out_codes, void** out_vals /*[naggs][G]*/,
                     uint64_t** out_counts /*[naggs][G]*/, int64_t* out_groups);

/* ---- shuffle support --------------------------------------- */

/* replaces dask's hash-repartition "tasks" shuffle split
 * (dask_sql/__init__.py:16, conftest.py:17; inside dd.merge/groupby):
 * bucket rows by mix64(code) % nbuckets, stable within bucket. Emits the
 * per-bucket-contiguous ORDERED selection vector into caller-allocated
 * out_sel u32[n] and bucket row offsets into out_offsets i64[nbuckets+1].
 * The host gathers each bucket's columns into torch-allocated staging
 * buffers and exchanges them with torch.distributed all_to_all (RCCL/xGMI). */
int dsx_partition(DsxCtx* ctx, const uint64_t* codes, const uint8_t* validity,
                  int64_t n, int nbuckets, uint32_t* out_sel,
                  int64_t* out_offsets);

/* TEST INFRASTRUCTURE: emit the hipRTC expression-evaluator source the
 * JIT generates for `prog` (host-only, no GPU) so CPU tests can compile it
 * with gcc and differential-test codegen semantics. Returns 1 when the
 * expression kind is double, 0 for i64, <0 on error. */
int dsx_jit_expr_source(const DsxInstr* prog, int prog_len,
                        const int32_t* dtypes, const uint8_t* has_validity,
                        int ncols, char* buf, int64_t cap);

/* TEST INFRASTRUCTURE: emit the JIT key-pack source for a key spec
 * (host-only, no GPU) for the CPU pack-semantics differential tests. */
int dsx_jit_pack_source(const DsxKeySpec* keys, int nkeys,
                        const int32_t* dtypes, const uint8_t* has_validity,
                        int ncols, char* buf, int64_t cap);

#ifdef __cplusplus
}
#endif
#endif /* DSXHIP_H */
