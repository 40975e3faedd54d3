/* ballista_gpu.h — C ABI of the MI355X-native Ballista stage-executor
 * library (libballista_gpu.so).
 *
 * This is the device boundary of SURVEY.md §8(b), seam 3: the thin
 * `extern "C"` FFI a Rust `GpuExecutionEngine` (implementing Ballista's
 * `ExecutionEngine` trait, ballista/executor/src/execution_engine.rs:53-103,
 * injected via `Executor::new` arg 8 / `ExecutorProcessConfig.
 * override_execution_engine`, executor_process.rs:319) binds to run the
 * executor-side physical-operator hot path on a gfx950 GPU.  The Rust-side
 * binding stub a maintainer would add is shown in INTEGRATION.md.
 *
 * Conventions:
 *  - All functions return BG_OK (0) or a negative bg_status code;
 *    bg_last_error() returns a thread-local message for the last failure.
 *  - Column buffers are Arrow C-Data-Interface-style: raw data pointer +
 *    optional validity bitmap (LSB bit order) + length.  Pointers named d_*
 *    are DEVICE pointers owned by the caller (allocated via bg_malloc);
 *    the library never frees caller memory.
 *  - There is no CPU fallback anywhere behind this ABI: every compute entry
 *    point requires an initialised HIP device and fails loudly otherwise.
 *
 * Reference interfaces each entry point replaces (file:line under
 * /root/reference):
 *  - bg_hash_columns / bg_partition_ids / bg_partition_indices:
 *      compute_partition_indices,
 *      ballista/core/src/execution_plans/sort_shuffle/writer.rs:1259-1279
 *      (evaluate key exprs -> create_hashes(REPARTITION_RANDOM_STATE) ->
 *       h % K -> per-partition row-index lists).
 *  - bg_eval_predicates / bg_mask_to_indices / bg_gather:
 *      DataFusion 55 FilterExec predicate eval + filter compaction and the
 *      sort-shuffle writer's interleave-gather
 *      (sort_shuffle/partitioned_batch_iterator.rs; SURVEY.md §8a rows 1,5).
 *  - bg_hash_repartition: the device half of
 *      SortShuffleWriterExec::execute_shuffle_write (writer.rs:564-753) —
 *      bucket rows by partition and materialise partition-major column
 *      buffers; IPC encode + file write stay host-side
 *      (write_task_consolidated writer.rs:810-895 + index.rs:21-33).
 *  - bg_q6_agg: the fused Filter+Partial-Aggregate stage of TPC-H q6
 *      (plan golden scheduler/tests/tpch_plan_stability/approved/q6.txt;
 *       SUM(Decimal128) exact i128 per DataFusion AggregateExec).
 *  - bg_q1_agg: the fused Filter+Partial-Aggregate stage of TPC-H q1
 *      (approved/q1.txt; grouped sums/counts, groups dictionary-encoded).
 */

#ifndef BALLISTA_GPU_H
#define BALLISTA_GPU_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- status ---- */
typedef enum {
  BG_OK = 0,
  BG_ERR_HIP = -1,        /* HIP runtime failure (see bg_last_error) */
  BG_ERR_NO_GPU = -2,     /* no usable gfx950 device / not initialised */
  BG_ERR_INVALID = -3,    /* bad argument */
  BG_ERR_UNSUPPORTED = -4 /* dtype/op not yet accelerated */
} bg_status;

const char* bg_last_error(void);
int bg_version(void);
/* sha256 over the library's own sources, injected at compile time by the
 * build entry; "unverified-local-build" when compiled without it.  A CPU
 * test compares it against a fresh hash of the committed sources so a
 * stale committed binary is detected. */
const char* bg_source_hash(void);
/* duration (ms) of the most recent fused-aggregate kernel launch, measured
 * with hipEvents on the launch stream — roofline evidence for bench.py */
double bg_last_kernel_ms(void);

/* ---- device/session ---- */
int bg_init(int device_ordinal); /* fails loudly when no GPU is present */
int bg_device_count(int* out);
int bg_synchronize(void);

/* ---- device memory (caller-owned) ---- */
int bg_malloc(uint64_t bytes, void** d_ptr); /* pooled; see bg_pool_trim */
int bg_free(void* d_ptr);
int bg_memset(void* d_ptr, int value, uint64_t bytes);
int bg_memcpy_h2d(void* d_dst, const void* h_src, uint64_t bytes);
int bg_memcpy_d2h(void* h_dst, const void* d_src, uint64_t bytes);
int bg_memcpy_dtod(void* d_dst, const void* d_src, uint64_t bytes);
/* release the allocator pool's cached device memory back to HIP */
int bg_pool_trim(void);

/* ---- columns ---- */
typedef enum {
  BG_DT_INT32 = 1,
  BG_DT_INT64 = 2,
  BG_DT_DATE32 = 3,      /* Arrow Date32: i32 days since epoch */
  BG_DT_DECIMAL128 = 4,  /* Arrow Decimal128: 16-byte LE two's complement */
  BG_DT_DICT8 = 5,       /* dictionary codes as u8 (small Utf8 dictionaries) */
  BG_DT_UTF8 = 6,        /* Arrow Utf8: i32 offsets (d_offsets) + byte data */
  BG_DT_FLOAT64 = 7,     /* f64; predicates/aggregates at 1e-6 rel tolerance */
} bg_dtype;

typedef struct {
  int32_t dtype;          /* bg_dtype */
  int32_t precision;      /* decimals only */
  int32_t scale;          /* decimals only */
  int32_t _pad;
  const void* d_data;     /* device pointer; UTF8: byte data buffer */
  const uint8_t* d_validity; /* device pointer or NULL (all valid); LSB bits */
  const int32_t* d_offsets;  /* UTF8 only: n+1 Arrow offsets; else NULL */
  int64_t len;
} bg_column;

/* ---- predicates (FilterExec subset; ops shared with the oracle) ---- */
typedef enum {
  BG_PRED_GE_LT = 0,   /* lo <= x <  hi */
  BG_PRED_BETWEEN = 1, /* lo <= x <= hi */
  BG_PRED_LT = 2,      /* x <  hi */
  BG_PRED_EQ = 3,      /* x == lo */
  BG_PRED_GT = 4,      /* x >  lo */
} bg_pred_op;

typedef struct {
  int32_t column; /* index into the cols array */
  int32_t op;     /* bg_pred_op */
  /* bounds as i128 split into (lo64, hi64); narrower dtypes use the low part */
  int64_t lo_lo; int64_t lo_hi;
  int64_t hi_lo; int64_t hi_hi;
} bg_pred;

/* AND-fold `npreds` predicates over `cols` into an Arrow LSB bitmask
 * d_mask (ceil(n/64)*8 bytes, 8-byte aligned). Null input => false. */
int bg_eval_predicates(const bg_column* cols, int32_t ncols,
                       const bg_pred* preds, int32_t npreds,
                       int64_t n, uint8_t* d_mask);

/* Stable compaction: selection bitmask -> ascending row indices.
 * d_indices capacity n; *out_count receives the selected count. */
int bg_mask_to_indices(const uint8_t* d_mask, int64_t n, uint32_t* d_indices,
                       int64_t* out_count);

/* Row gather (take): d_dst[i] = d_src[d_idx[i]], elem_size in {1,2,4,8,16}. */
int bg_gather(const void* d_src, int64_t elem_size, const uint32_t* d_idx,
              int64_t m, void* d_dst);

/* Variable-length (Utf8/Binary) gather: m rows by index -> Arrow i32
 * offsets (m+1) + packed bytes (the writer's take/interleave for string
 * payload columns). */
int bg_gather_varlen(const void* d_src_data, const int32_t* d_src_offsets,
                     const uint32_t* d_idx, int64_t m,
                     int32_t* d_out_offsets, void* d_out_data,
                     int64_t out_data_cap, int64_t* out_total_bytes);

/* Arrow i32 offset rebase (writer-side partition slicing). */
int bg_sub_i32(const void* d_src, int64_t n, int32_t sub, void* d_out);

/* Validity-bitmap gather: out bit i = valid[idx[i]] (Arrow LSB order;
 * d_out_bits holds ceil(m/64)*8 bytes) — take for null-carrying columns. */
int bg_gather_bits(const uint8_t* d_valid, const uint32_t* d_idx, int64_t m,
                   uint8_t* d_out_bits);

/* LIKE / NOT LIKE over a Utf8 column, ANDed into an existing Arrow LSB
 * bitmask (q9/q13/q14/q16-class predicates: in-order literal fragments
 * with optional start/end anchors — the forms the reference's
 * approved plans use; no '_' wildcards there). */
int bg_eval_like(const bg_column* col, const char* const* terms,
                 const int32_t* term_lens, int32_t nterms,
                 int32_t anchor_prefix, int32_t anchor_suffix,
                 int32_t negate, int64_t n, uint8_t* d_mask);

/* CASE WHEN mask THEN a ELSE b: elementwise select (esz 1..16). */
int bg_select(const uint8_t* d_mask, const void* d_a, const void* d_b,
              int64_t esz, int64_t n, void* d_out);
/* constant fill (literal CASE branches): value as i128 (lo, hi). */
int bg_fill_const(void* d_out, int64_t n, int64_t esz, int64_t lo,
                  int64_t hi);

/* x IN {v0..v15} over integer/dict columns, ANDed into an existing
 * bitmask (q12/q19-class IN-lists). */
int bg_eval_in(const bg_column* col, const int64_t* values, int32_t nvalues,
               int64_t n, uint8_t* d_mask);
/* out = a OR b (disjunctive predicate groups, q19's OR-of-ANDs). */
int bg_bitmap_or(const uint8_t* d_a, const uint8_t* d_b, int64_t nbits,
                 uint8_t* d_out);

/* ---- hash repartition (SortShuffleWriterExec device half) ---- */

/* create_hashes restatement over the key columns (bg_ahash.h; parity
 * unpinned for the ahash seed constants — SURVEY.md §8c). */
int bg_hash_columns(const bg_column* key_cols, int32_t nkeys, int64_t n,
                    uint64_t* d_hashes);

/* pid[i] = hash[i] % k  (writer.rs:1274-1276) */
int bg_partition_ids(const uint64_t* d_hashes, int64_t n, uint32_t k,
                     uint32_t* d_pids);

/* Stable multi-split: partition-major row-index lists, ascending row order
 * inside each partition — the device equivalent of the Vec<Vec<u32>> that
 * compute_partition_indices returns.  d_indices: n entries; d_offsets: k+1
 * exclusive prefix (offsets[k] == n).  k <= 4096. */
int bg_partition_indices(const uint32_t* d_pids, int64_t n, uint32_t k,
                         uint32_t* d_indices, int64_t* d_offsets);

/* As bg_partition_indices, also emitting the inverse permutation
 * d_rank[row] = output position (coalesced write) — feeds
 * bg_scatter_rows so payload materialisation reads inputs sequentially. */
int bg_partition_indices_ex(const uint32_t* d_pids, int64_t n, uint32_t k,
                            uint32_t* d_indices, int64_t* d_offsets,
                            uint32_t* d_rank);

/* Scatter-materialise: d_dst[d_rank[i]] = d_src[i] (sequential reads,
 * partition-major writes — the write-side dual of bg_gather, avoiding the
 * gather's k-fold read amplification on permuted input). */
int bg_scatter_rows(const void* d_src, int64_t elem_size,
                    const uint32_t* d_rank, int64_t n, void* d_dst);

/* Fused convenience: hash + pids + stable split + gather every payload
 * column partition-major.  d_out[c] must hold len*elem_size(c) bytes; rows
 * of partition p for column c live at [offsets[p]*esz, offsets[p+1]*esz).
 * offsets (k+1, device) and the permutation d_indices are also returned so
 * the host can slice buffers for IPC encoding. */
int bg_hash_repartition(const bg_column* key_cols, int32_t nkeys,
                        const bg_column* payload_cols, int32_t ncols,
                        int64_t n, uint32_t k,
                        uint32_t* d_indices, int64_t* d_offsets,
                        void** d_out /* ncols device pointers */);

/* ---- HashJoinExec build/probe (INNER equi-join, Int64 keys) ----
 * Replaces DataFusion 55 HashJoinExec's build/probe arithmetic for
 * partitioned hash joins (planned when
 * `prefer_hash_join=true`, core/src/extension.rs:850-858 + the opt-in test
 * client/tests/context_checks.rs:1034-1063).  Contract: the emitted
 * (probe_idx, build_idx) pair MULTISET equals the reference's inner-join
 * result set; pairs are probe-major (ascending probe_idx, exact per-row
 * offsets from the count phase); chain order within one probe row is
 * unspecified, as in the reference (SQL does not pin intra-row order). */
int bg_hashjoin_build(const bg_column* build_keys, int64_t n_build,
                      void** out_handle);
int bg_hashjoin_probe_count(void* handle, const bg_column* probe_keys,
                            int64_t n_probe, int64_t* out_matches);
int bg_hashjoin_probe_fill(void* handle, const bg_column* probe_keys,
                           int64_t n_probe, uint32_t* d_out_probe,
                           uint32_t* d_out_build);
int bg_hashjoin_free(void* handle);

/* ---- generalized hash join (round 2): multi-column keys over
 * Int64/Int32/Date32/Decimal128/Utf8/dict8 and the probe-side join types
 * of DataFusion's HashJoinExec (q2/q9-class shapes).  Nodes store
 * {hash, next}; equality is the cross-table key compare.  Join null
 * semantics (null_equals_null=false): a row with ANY null key matches
 * nothing — excluded by INNER/SEMI, emitted by ANTI/OUTER_PROBE with
 * build id BG_JOIN_NULL_IDX.  The build-key device buffers must outlive
 * the handle (nodes reference them for compares). */
#define BG_JOIN_INNER 0
#define BG_JOIN_SEMI 1        /* probe rows with >=1 match, emitted once */
#define BG_JOIN_ANTI 2        /* probe rows with no match */
#define BG_JOIN_OUTER_PROBE 3 /* matches + unmatched probe rows */
#define BG_JOIN_NULL_IDX 0xFFFFFFFFu
int bg_hashjoin_build2(const bg_column* keys, int32_t nkeys, int64_t n,
                       void** out_handle);
int bg_hashjoin_probe_count2(void* handle, const bg_column* keys,
                             int32_t nkeys, int64_t n, int32_t join_type,
                             int64_t* out_matches);
int bg_hashjoin_probe_fill2(void* handle, const bg_column* keys,
                            int32_t nkeys, int64_t n, int32_t join_type,
                            uint32_t* d_out_probe, uint32_t* d_out_build);
int bg_hashjoin_free2(void* handle);

/* Split an index vector carrying BG_JOIN_NULL_IDX sentinels (probe-outer
 * build side) into a clamped gather-safe vector + the validity bitmap of
 * non-sentinel slots. */
int bg_idx_sentinel(const uint32_t* d_idx, int64_t m, uint32_t* d_clamped,
                    uint8_t* d_valid_bits);
/* out = a AND b over ceil(nbits/8) bytes (combine validity bitmaps). */
int bg_bitmap_and(const uint8_t* d_a, const uint8_t* d_b, int64_t nbits,
                  uint8_t* d_out);

/* ---- general hash group-by (AggregateExec Partial/Single) ----
 * Arbitrary group cardinality (q3-class: millions of groups).  Aggregate
 * ops: 0 = SUM over Decimal128 (exact i128), 1 = SUM over Int64 (exact,
 * accumulated in i128).  COUNT(*) is always produced.  d_mask (optional,
 * Arrow LSB bitmask words) pre-filters rows (fused Filter+Aggregate).
 * Outputs are dense groups in ascending hash-slot order (deterministic for
 * a fixed max_groups): d_first_row[g] = first input row carrying the
 * group's key (gather key values from it), d_acc_out = i128 LE per
 * (group, agg), d_counts_out = i64 per group. */
#define BG_AGG_OP_SUM_DEC128 0
#define BG_AGG_OP_SUM_I64 1
#define BG_AGG_OP_MIN_I64 2 /* acc = order-preserving u64; host decodes */
#define BG_AGG_OP_MAX_I64 3
#define BG_AGG_OP_SUM_F64 4
#define BG_AGG_OP_MIN_F64 5 /* acc = totally-ordered u64 (IEEE sign flip) */
#define BG_AGG_OP_MAX_F64 6
int bg_hashagg(const bg_column* key_cols, int32_t nkeys,
               const bg_column* agg_cols, const int32_t* agg_ops,
               int32_t naggs, const uint8_t* d_mask, int64_t n,
               int64_t max_groups, uint32_t* d_first_row,
               uint8_t* d_acc_out /* max_groups*naggs*16 */,
               int64_t* d_counts_out /* max_groups */,
               int64_t* out_ngroups);

/* As bg_hashagg, plus d_nncnt_out (max_groups*naggs i64): the per-
 * (group, aggregate) count of NON-NULL inputs.  NULL aggregate inputs are
 * always skipped (SQL semantics; the reference's accumulators in
 * datafusion/functions-aggregate treat null as no contribution); a group
 * whose inputs were all NULL has nncnt 0 and its SUM/MIN/MAX is NULL —
 * the accumulator value alone cannot encode that.  NULL group keys hash
 * as "no contribution" (hash_utils create_hashes) and group together
 * (NULL == NULL in GROUP BY). */
int bg_hashagg2(const bg_column* key_cols, int32_t nkeys,
                const bg_column* agg_cols, const int32_t* agg_ops,
                int32_t naggs, const uint8_t* d_mask, int64_t n,
                int64_t max_groups, uint32_t* d_first_row,
                uint8_t* d_acc_out, int64_t* d_counts_out,
                int64_t* d_nncnt_out, int64_t* out_ngroups);

/* ---- ProjectionExec expression subset: Decimal128 arithmetic ----
 * ops: 0 a*b, 1 a+b, 2 a-b, 3 lit-a, 4 a*lit, 5 a+lit (exact i128;
 * lit as (lo,hi) i128 halves).  d_out: n x 16 B. */
int bg_project_dec128(int32_t op, const bg_column* a, const bg_column* b,
                      int64_t lit_lo, int64_t lit_hi, int64_t n, void* d_out);

/* Multi-expression Decimal128 projection in one pass: stack bytecode
 * (PUSH_COL/PUSH_LIT/MUL/ADD/SUB) over <=8 input columns, <=6 output
 * dec128 columns; each input read once, intermediates in registers (the
 * q1-class expression chains cost ~77 GB of materialised intermediates
 * at SF100 without this). */
int bg_project_dec128_multi(const bg_column* cols, int32_t ncols,
                            const int32_t* ops, const int32_t* args,
                            int32_t nops, const int32_t* expr_end,
                            int32_t nexprs, const int64_t* lit_lo,
                            const int64_t* lit_hi, int32_t nlits, int64_t n,
                            void* const* d_outs);

/* ---- SortExec (stable multi-column ORDER BY) ----
 * d_perm (u32[n]) receives the stable row permutation realising ORDER BY
 * key_cols[0] [DESC], key_cols[1] [DESC], ...; Top-K = first K entries.
 * Keys: INT64/INT32/DATE32 (round 1). */
int bg_sort_rows(const bg_column* key_cols, const int32_t* descending,
                 int32_t nkeys, int64_t n, uint32_t* d_perm);

/* As bg_sort_rows with explicit per-key null ordering (1 = NULLS FIRST).
 * bg_sort_rows defaults to the SQL convention (ASC -> NULLS LAST,
 * DESC -> NULLS FIRST); null keys sort as one stable group and never
 * scramble by their undefined payload bytes. */
int bg_sort_rows2(const bg_column* key_cols, const int32_t* descending,
                  const int32_t* nulls_first, int32_t nkeys, int64_t n,
                  uint32_t* d_perm);

/* ---- SortMergeJoinExec (INNER, key-sorted Int64 inputs) ----
 * The reference's DEFAULT partitioned join (prefer_hash_join=false,
 * extension.rs:850-858; every approved/q*.txt join stage is SMJ).  Inputs
 * are key-sorted (bg_sort_rows); emits positions into the sorted orders,
 * probe-major with build ascending (fully deterministic).  Call once with
 * d_out_* NULL to size the output, then again with buffers. */
int bg_merge_join(const int64_t* d_build_sorted, int64_t nb,
                  const int64_t* d_probe_sorted, int64_t np,
                  int64_t* out_matches, uint32_t* d_out_probe,
                  uint32_t* d_out_build);

/* ---- GPU Parquet decode, stage 1: Snappy page decompression ----
 * (SURVEY.md §8f row 1).  One device thread decodes one independently-
 * compressed page; thousands of pages decode concurrently.  h_pages is a
 * host array of {src, dst, src_len, dst_cap} with DEVICE src/dst pointers;
 * h_out_lens receives decompressed lengths (-1 = malformed page). */
typedef struct {
  const void* d_src;
  void* d_dst;
  int64_t src_len;
  int64_t dst_cap;
} bg_snappy_page;
int bg_snappy_decompress(const void* h_pages /* bg_snappy_page[npages] */,
                         int64_t npages, int64_t* h_out_lens);

/* Parquet data-page extraction (PLAIN, no nulls): validates the def-level
 * block (optional columns) on device, copies the value bytes into the
 * column buffer at dst_byte_off; flba_reverse flips 16-B big-endian
 * decimals to Arrow LE.  Fails loudly on nulls/V2/dictionary pages. */
int bg_page_extract(const void* d_page, int64_t page_len, void* d_out,
                    int64_t dst_byte_off, int64_t nvals, int64_t src_esz,
                    int32_t has_def, int32_t flba_reverse);

/* Parquet dictionary-index expansion (PLAIN_DICTIONARY/RLE_DICTIONARY
 * data pages): one device thread expands one page's RLE/bit-packed index
 * block to u32 indices; materialise with bg_gather from the PLAIN-decoded
 * dictionary. */
int bg_dict_indices(const void* d_page, int64_t page_len, int64_t nvals,
                    int32_t has_def, uint32_t* d_out_idx);

/* Batched forms: the whole column's page list in one launch (chunks hold
 * few pages; per-page host loops starve the GPU). */
typedef struct {
  const void* d_page;
  void* d_out;
  int64_t page_len;
  int64_t nvals;
  int64_t src_esz;
  int32_t has_def;      /* 0 none, 1 validate-all-present, 2 nullable */
  int32_t flba_reverse;
  const uint32_t* d_vidx;      /* mode 2: slot -> value index, ~0u = NULL */
  const int64_t* d_n_present;  /* mode 2: values in this page */
} bg_page_extract_job;
int bg_page_extract_batch(const void* h_jobs, int64_t njobs);

typedef struct {
  const void* d_page;
  uint32_t* d_out_idx;
  int64_t page_len;
  int64_t nvals;
  int32_t has_def;      /* 0 none, 1 validate-all-present, 2 nullable */
  int32_t _pad;
  const uint32_t* d_vidx;      /* mode 2 */
  const int64_t* d_n_present;  /* mode 2 */
  uint32_t* d_dense;           /* mode 2: scratch, >= n_present u32 */
} bg_dict_indices_job;
int bg_dict_indices_batch(const void* h_jobs, int64_t njobs);

/* OPTIONAL-column definition levels (max_def=1; the RLE/bit-packed hybrid
 * of the parquet spec, as decoded by the reference's parquet crate
 * rle.rs): per data page, decode [u32 len][levels] into the column's
 * Arrow validity bitmap (atomicOr into u32 words at bit_off — page
 * boundaries are not byte-aligned), the slot->value-index map vidx
 * (~0u = NULL), and n_present.  Feed vidx/n_present to the mode-2
 * extract/dict jobs above. */
typedef struct {
  const void* d_page;    /* page start ([u32 dlen][levels][values...]) */
  uint32_t* d_vidx;      /* u32[nvals] slice for this page */
  uint32_t* d_valid_out; /* column validity bitmap (u32 words, zeroed) */
  int64_t page_len;
  int64_t nvals;
  int64_t bit_off;       /* absolute bit position of this page's slot 0 */
  int64_t* d_n_present;
} bg_def_levels_job;
int bg_def_levels_batch(const void* h_jobs, int64_t njobs);

/* LIST columns (repetition levels, max_rep == 1; parquet-format.md
 * "Nested Encoding" / the reference's arrow-rs list reader).  A V1 page
 * body is [u32 rlen][rep][u32 dlen][def][values].  pass 1 fills
 * d_counts[4] = {rows, entries, present, rlen} per page; the host
 * prefix-sums bases and re-points the UNCHANGED mode-2 extract/dict
 * jobs at page + 4 + rlen.  pass 2 emits per-row entry counts (list
 * offsets = their prefix sum), the list/element validity bitmaps and
 * the page-local value index per entry.  Rows must not span pages. */
typedef struct {
  const void* d_page;   /* [u32 rlen][rep][u32 dlen][def][values] */
  int64_t page_len;
  int64_t nslots;       /* level entries (page header num_values) */
  int32_t max_def;
  int32_t def_entry;    /* min def meaning an element slot exists */
  int32_t def_valid;    /* min def meaning the LIST is non-null */
  int32_t _pad;
  int64_t row_base;     /* pass 2: column-global bases */
  int64_t entry_base;
  int64_t* d_counts;    /* pass 1 */
  int32_t* d_row_sizes; /* pass 2 (column-global) */
  uint32_t* d_list_valid;
  uint32_t* d_elem_valid;
  uint32_t* d_vidx;     /* pass 2: PAGE-LOCAL slice */
} bg_list_levels_job;
int bg_list_levels_batch(const void* h_jobs, int64_t njobs, int32_t pass);

/* BYTE_ARRAY (Utf8/Binary) PLAIN pages (parquet spec PLAIN: [u32 len]
 * [bytes] per value; the reference's decoding.rs PlainDecoder):
 * per page, record each slot's length + absolute device source address
 * (NULL slots length 0); then ONE column-wide bg_ba_materialize scans the
 * lengths into Arrow i32 offsets and copies every slot's bytes.  Dict-
 * coded string pages bridge with bg_ba_from_dict (lengths/addresses from
 * the PLAIN-decoded dictionary through the expanded indices). */
typedef struct {
  const void* d_page;
  int64_t* d_lens_out;     /* i64[nvals] slice */
  int64_t* d_srcaddr_out;  /* i64[nvals] slice */
  int64_t page_len;
  int64_t nvals;
  int32_t has_def; /* 0 none, 2 nullable */
  int32_t _pad;
  const uint32_t* d_vidx;
  const int64_t* d_n_present;
} bg_ba_page_job;
int bg_ba_extract_batch(const void* h_jobs, int64_t njobs);
int bg_ba_from_dict(const uint32_t* d_idx, const int32_t* d_doffs,
                    const void* d_ddata, const uint32_t* d_vidx, int64_t n,
                    int64_t* d_lens_out, int64_t* d_srcaddr_out);
int bg_ba_materialize(const int64_t* d_lens, const int64_t* d_srcaddr,
                      int64_t n, int32_t* d_offs32 /* n+1 */,
                      uint8_t* d_data, int64_t data_cap, int64_t* out_total);

/* DELTA_BINARY_PACKED (encoding 5) INT32/INT64 pages — the reference's
 * parquet-rs DataPageV2 writer default for integers (spec Encodings.md
 * "Delta Encoding"; decoding.rs DeltaBitPackDecoder).  Lane-0 serial per
 * page (the prefix chain is sequential), pages concurrent across waves;
 * nullable slots scatter through the def-level vidx map. */
typedef struct {
  const void* d_page;
  void* d_out;           /* i32/i64 column slice */
  int64_t page_len;
  int64_t nvals;
  int64_t esz;           /* 4 or 8 */
  int32_t has_def;       /* 0 none, 2 nullable */
  int32_t _pad;
  const uint32_t* d_vidx;
  const int64_t* d_n_present;
} bg_delta_bp_job;
int bg_delta_bp_batch(const void* h_jobs, int64_t njobs);

/* DELTA_LENGTH_BYTE_ARRAY (6) / DELTA_BYTE_ARRAY (7) string pages — the
 * reference's parquet-rs V2 writer defaults for BYTE_ARRAY (Encodings.md;
 * decoding.rs DeltaLengthByteArrayDecoder / DeltaByteArrayDecoder).
 * pass 1 fills slot lengths (+ source addresses for enc 6; 0 for enc 7);
 * after bg_ba_materialize lays out the column, pass 2 (enc 7 only)
 * rebuilds each string from its predecessor's prefix + its suffix. */
typedef struct {
  const void* d_page;
  int64_t* d_lens_out;
  int64_t* d_srcaddr_out;
  const int32_t* d_offs32;  /* pass 2 */
  uint8_t* d_data_out;      /* pass 2 */
  int64_t page_len;
  int64_t nvals;
  int32_t has_def;
  int32_t enc; /* 6 or 7 */
  const uint32_t* d_vidx;
  const int64_t* d_n_present;
} bg_delta_ba_job;
int bg_delta_ba_batch(const void* h_jobs, int64_t njobs, int32_t pass);

/* BYTE_STREAM_SPLIT (encoding 9, fixed-width): parallel byte transpose
 * of the k per-byte streams back into values (bg_page_extract_job fields;
 * flba_reverse unused). */
int bg_bss_batch(const void* h_jobs, int64_t njobs);

/* Device LZ4 block compression (the GPU shuffle codec's compress half,
 * SURVEY.md §8f row 3): 64 KiB blocks, one wave per block.  d_out_slots
 * holds nblocks slots of 65544 B; h_block_sizes[i] = compressed size, or
 * negative (-usize) when the block is stored raw.  Host glue frames the
 * blocks (constant header 04224d18 40 40 c0 + [u32 size] blocks + end
 * mark) into the LZ4_FRAME streams Arrow IPC carries. */
#define BG_LZ4_BLOCK 65536
#define BG_LZ4_SLOT_STRIDE 65544
int bg_lz4_compress(const void* d_src, int64_t len, void* d_out_slots,
                    int64_t* h_block_sizes, int64_t* out_nblocks);
/* flat batched compression over many buffers' blocks in one launch */
typedef struct {
  const void* d_src;
  void* d_dst_slot;
  int32_t blen;
  int32_t _pad;
} bg_lz4_block_job;
int bg_lz4_compress_flat(const void* h_jobs, int64_t njobs,
                         int64_t* h_block_sizes);
/* LZ4-frame DECODE on device (shuffle-read ingest; mirror of the
 * compressor).  h_frames entries point at frame magic; h_out_lens[i] =
 * decompressed bytes or -1 on malformed input. */
typedef struct {
  const void* d_src;
  void* d_dst;
  int64_t src_len;
  int64_t dst_cap;
} bg_lz4_frame;
int bg_lz4_decompress(const void* h_frames, int64_t nframes,
                      int64_t* h_out_lens);
/* assemble [u32 size][block] sequences on device at precomputed offsets */
typedef struct {
  const void* d_src;
  void* d_dst;        /* at the 4-byte size word */
  int64_t nbytes;
  uint32_t size_word; /* high bit = stored block */
  uint32_t _pad;
} bg_pack_job;
int bg_pack_blocks(const void* h_jobs, int64_t njobs);

/* Fused repartition materialiser (k <= 64, <= 4 fixed-width payload
 * columns): one pass re-hashes keys, ranks rows per partition with
 * ballots, and write-combines k scattered payload streams through
 * per-partition LDS tiles (64-row flushes).  Same outputs and stable
 * order as bg_hash_repartition. */
int bg_hash_repartition_fused(const bg_column* key_cols, int32_t nkeys,
                              const bg_column* payload_cols, int32_t ncols,
                              int64_t n, uint32_t k, uint32_t* d_indices,
                              int64_t* d_offsets, uint32_t* d_rank,
                              void** d_out);

/* ---- stage-interpreter support (bg_execute_stage internals that are
 * also useful standalone) ---- */

/* Place nbits of an Arrow LSB bitmap at an arbitrary bit offset of a
 * destination bitmap (device shuffle-read: batch validity bitmaps land at
 * non-byte-aligned row cursors).  Boundary words use atomics. */
int bg_bitcopy(const uint8_t* d_src_bits, int64_t nbits, uint8_t* d_dst_bits,
               int64_t dst_bit_off);

/* Materialize bg_hashagg accumulator records into an Arrow column:
 * acc_stride in bytes between consecutive groups' records for this
 * aggregate; op = BG_AGG_OP_*; d_out element size 16 (SUM_DEC128) or 8.
 * With d_nncnt (stride bytes), groups whose non-null input count is 0 get
 * their bit cleared in d_valid_out (SQL: SUM/MIN/MAX of all-NULL = NULL). */
int bg_agg_materialize(const void* d_acc, int64_t acc_stride, int32_t op,
                       int64_t ngroups, void* d_out, const int64_t* d_nncnt,
                       int64_t nncnt_stride, uint8_t* d_valid_out);

/* Strided i64 copy (COUNT materialisation from interleaved agg records). */
int bg_copy_i64_strided(const void* d_src, int64_t stride, int64_t n,
                        void* d_out);

/* AVG finalisation (AggregateExec Final for avg):
 * decimal — DataFusion's decimal AvgAccumulator: target scale = input
 * scale + 4, out = round-half-away-from-zero(sum * 10^scale_shift / count),
 * exact i128; f64 — IEEE sum/count.  count==0 clears the group's valid
 * bit. */
int bg_avg_finalize(const void* d_sum_acc, int64_t acc_stride, int32_t is_f64,
                    const void* d_cnt, int64_t cnt_stride,
                    int32_t scale_shift, int64_t ngroups, void* d_out,
                    uint8_t* d_valid_out);

/* ---- stage interpreter (the product entry of SURVEY.md §8b seam 1) ----
 *
 * bg_execute_stage executes ONE whole query stage — the GPU analogue of
 * QueryStageExecutor::execute_query_stage (executor/src/execution_engine.
 * rs:78-103, default impl :127-212): operators are sequenced by a plan
 * tree, key EXPRESSIONS are evaluated before hashing (sort_shuffle/
 * writer.rs:1265), and the root writer produces the reference's exact
 * shuffle file bytes + ShuffleWritePartition summaries (proto :779-791).
 *
 * plan_json is a faithful JSON restatement of the decoded TaskDefinition
 * plan (the Rust host decodes task.plan with datafusion-proto exactly as
 * execution_loop.rs:364-367 does, then serialises the operator tree to
 * this schema — see INTEGRATION.md "Stage plan JSON").  Operators:
 * scan / filter / project / hash_join / hash_aggregate / sort /
 * sort_shuffle_write / passthrough_write / collect.
 *
 * On success *out_json receives a malloc'd result document:
 *   {"partitions": [{"partition_id", "path", "num_batches", "num_rows",
 *     "num_bytes"}...], "metrics": {...}, "rows": [...] (collect roots)}
 * Free it with bg_stage_free.  On failure returns a bg_status and
 * bg_last_error() describes the offending plan node. */
int bg_execute_stage(const char* plan_json, char** out_json);
void bg_stage_free(char* p);

/* Parse + type-check a stage plan without touching the GPU (host-side
 * validation; CPU tests cover the plan grammar through this). */
int bg_stage_validate(const char* plan_json, char** out_json);

/* Register a device-resident input table for "device" scan sources
 * (bench/tests feed synthetic in-HBM tables; the Rust host feeds decoded
 * ShuffleReaderExec batches the same way).  The caller owns the column
 * buffers; they must outlive the stage. */
int bg_stage_register_table(const char* name, const bg_column* cols,
                            const char* const* col_names, int32_t ncols,
                            int64_t n_rows);
int bg_stage_unregister_table(const char* name);

/* Deterministic synthetic fill (splitmix64 per element) for the
 * no-Python C++ examples/bench — test/bench infrastructure, not a query
 * path.  mode: 0 int64, 1 int32, 2 Decimal128 (LE pair).  Values uniform
 * in [lo, hi). */
int bg_fill_rand(void* d_out, int64_t n, uint64_t seed, int64_t lo,
                 int64_t hi, int32_t mode);

/* ---- fused filter+aggregate stages ---- */

/* TPC-H q6 stage 1 (scan+filter+aggregate, approved/q6.txt):
 *   sum(l_extendedprice * l_discount), count of qualifying rows.
 * shipdate: DATE32; discount/quantity/extendedprice: DECIMAL128.
 * Result is the exact i128 sum (scale 4). */
int bg_q6_agg(const bg_column* shipdate, const bg_column* discount,
              const bg_column* quantity, const bg_column* extendedprice,
              int32_t date_lo, int32_t date_hi, int64_t disc_lo,
              int64_t disc_hi, int64_t qty_lt,
              uint64_t* out_sum_lo, int64_t* out_sum_hi, int64_t* out_count);

/* TPC-H q1 stage 1 (filter + grouped partial aggregate, approved/q1.txt):
 * group = (rf_code<<4)|ls_code over u8 dictionary codes (<16 each);
 * per group: count + exact i128 sums of {qty, price, price*(100-disc),
 * price*(100-disc)*(100+tax), disc} — scales {2,2,4,6,2}.
 * h_counts: host i64[256]; h_sums: host bytes 256*5*16 (LE i128 each). */
int bg_q1_agg(const bg_column* rf, const bg_column* ls,
              const bg_column* quantity, const bg_column* extendedprice,
              const bg_column* discount, const bg_column* tax,
              const bg_column* shipdate, int32_t date_le,
              int64_t* h_counts, uint8_t* h_sums);

#ifdef __cplusplus
} /* extern "C" */
#endif

#endif /* BALLISTA_GPU_H */
