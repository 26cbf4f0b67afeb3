/* trino_gpu.h — C-ABI drop-in boundary for Trino's columnar operator hot path,
 * MI355X-native implementation (libtrino_gpu.so).
 *
 * Each entry point mirrors the io.trino SPI/operator interface a JNI shim would
 * bind (the reference has no native code; this is the surface LocalExecutionPlanner
 * -constructed operator factories would delegate to). Reference interfaces mirrored:
 *   - Page/Block:   core/trino-spi/src/main/java/io/trino/spi/Page.java:31,50-51
 *                   spi/block/Block.java:21-24 (ValueBlock|DictionaryBlock|RLE),
 *                   spi/block/LongArrayBlock.java:39-43 (flat values + valid bitmap),
 *                   spi/block/VariableWidthBlock.java:43-48, DictionaryBlock.java:55-58
 *   - SelectedPositions: operator/project/SelectedPositions.java:27-58
 *   - Operator:     operator/Operator.java:18-50
 *                   (needsInput()/addInput(Page)/getOutput()/finish()/isFinished())
 *   - OperatorFactory creation sites: sql/planner/LocalExecutionPlanner.java:2121
 *                   (scan+filter+project), :4080 (hash agg), :2966/:3017 (hash build),
 *                   OperatorFactories.java:24-47 (lookup join),
 *                   operator/output/PartitionedOutputOperator.java (partitioned output)
 * No torch types; plain pointers and sizes only. See INTEGRATION.md for the JNI
 * binding a Trino maintainer would add.
 */
#ifndef TRINO_GPU_H
#define TRINO_GPU_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- error codes (mirrors unchecked TrinoException at the boundary) ---- */
typedef enum {
    TG_OK = 0,
    TG_ERR_INVALID_ARG = 1,
    TG_ERR_NO_GPU = 2,        /* HIP device unavailable: the product path FAILS, no CPU fallback */
    TG_ERR_OOM = 3,
    TG_ERR_STATE = 4,         /* operator state machine violation */
    TG_ERR_HIP = 5,           /* underlying HIP error; see tg_last_error() */
    TG_ERR_UNSUPPORTED = 6
} tg_status;

const char* tg_last_error(void);
/* library version / build arch probe */
const char* tg_version(void);

/* ---- types (spi/type): the subset on the hot path ---- */
typedef enum {
    TG_BIGINT = 0,   /* i64  (LongArrayBlock)   */
    TG_INTEGER = 1,  /* i32  (IntArrayBlock)    */
    TG_SMALLINT = 2, /* i16  (ShortArrayBlock)  */
    TG_TINYINT = 3,  /* i8   (ByteArrayBlock)   */
    TG_DOUBLE = 4,   /* f64  (LongArrayBlock bits) */
    TG_DATE = 5,     /* i32 epoch days (IntArrayBlock) */
    TG_BOOLEAN = 6,  /* i8   (ByteArrayBlock)   */
    TG_VARCHAR = 7   /* VariableWidthBlock      */
} tg_type;

typedef enum { TG_BK_VALUE = 0, TG_BK_DICTIONARY = 1, TG_BK_RLE = 2 } tg_block_kind;

/* Flat column descriptor. Pointers may be HOST or DEVICE memory; `on_device`
 * says which. Host blocks are uploaded (flat memcpy of the backing arrays —
 * zero pivot) on addInput. */
typedef struct tg_block {
    int32_t type;               /* tg_type */
    int32_t kind;               /* tg_block_kind */
    int64_t position_count;
    int32_t on_device;          /* 0 = host pointers, 1 = device pointers */
    int32_t _pad;
    const void* data;           /* fixed-width values, or utf8 bytes for VARCHAR */
    const uint64_t* valid;      /* packed bitmap, bit=1 => valid (LongArrayBlock.valueIsValid); NULL => no nulls */
    const int32_t* offsets;     /* VARCHAR: N+1 offsets into data */
    const int32_t* ids;         /* DICTIONARY: position -> dictionary id */
    const struct tg_block* dictionary; /* DICTIONARY/RLE: the value block */
} tg_block;

typedef struct tg_page {
    int32_t channel_count;
    int64_t position_count;
    const tg_block* blocks;     /* array[channel_count] */
} tg_page;

/* SelectedPositions (operator/project/SelectedPositions.java:27-58) */
typedef struct tg_selected {
    int32_t is_list;            /* 0: range [offset, offset+size); 1: positions list */
    int32_t offset;
    int32_t size;
    const int32_t* positions;   /* when is_list */
} tg_selected;

/* ---- session = device context ---- */
typedef struct tg_session tg_session;
tg_status tg_session_create(int device_ordinal, tg_session** out);
void      tg_session_close(tg_session*);

/* ---- expression IR (replaces per-query bytecode gen, sql/gen/columnar/) ----
 * Postfix program over f64/i64 lanes; enough for TPC-H filter/project forms. */
typedef enum {
    TG_EXPR_COL = 0,       /* push column[arg0] value */
    TG_EXPR_CONST_F64 = 1, /* push f64 constant */
    TG_EXPR_CONST_I64 = 2, /* push i64 constant */
    TG_EXPR_ADD = 3, TG_EXPR_SUB = 4, TG_EXPR_MUL = 5, TG_EXPR_DIV = 6,
    /* comparisons produce boolean (filter roots) */
    TG_EXPR_LE = 7, TG_EXPR_LT = 8, TG_EXPR_GE = 9, TG_EXPR_GT = 10,
    TG_EXPR_EQ = 11, TG_EXPR_NE = 12,
    TG_EXPR_AND = 13, TG_EXPR_OR = 14, TG_EXPR_NOT = 15,
    TG_EXPR_BETWEEN = 16,  /* value, lo, hi on stack */
    TG_EXPR_IN_I64 = 17    /* arg0 = count, followed by imm list (host side) */
} tg_expr_op;

typedef struct tg_expr_inst {
    int32_t op;
    int32_t arg0;
    union { double f64; int64_t i64; } imm;
} tg_expr_inst;

typedef struct tg_expr { const tg_expr_inst* insts; int32_t count; } tg_expr;

/* ---- generic operator handle (operator/Operator.java:18-50) ---- */
typedef struct tg_operator tg_operator;
int       tg_operator_needs_input(tg_operator*);
tg_status tg_operator_add_input(tg_operator*, const tg_page*);
/* getOutput: fills out_page with DEVICE-resident blocks owned by the operator
 * until the next call; returns TG_OK with out_page->position_count==0 and
 * channel_count==0 when no output is ready. *finished set when drained. */
tg_status tg_operator_get_output(tg_operator*, tg_page* out_page, int* finished);
tg_status tg_operator_finish(tg_operator*);
void      tg_operator_close(tg_operator*);

/* ---- FilterAndProject (ScanFilterAndProjectOperator.java:66 /
 *      PageProcessor.java:102-138; filter = ColumnarFilter.java:42-52) ---- */
tg_status tg_filter_project_create(tg_session*,
    const tg_expr* filter,            /* NULL = select all */
    const tg_expr* projections,       /* array[n_proj] */
    const int32_t* proj_out_types,    /* tg_type per projection */
    int32_t n_proj,
    tg_operator** out);

/* Standalone columnar filter: predicate -> selection vector
 * (ColumnarFilter.filterPositionsRange/List). Synchronous helper used by tests. */
tg_status tg_filter_run(tg_session*, const tg_expr* filter, const tg_page* page,
                        const tg_selected* input_sel,
                        int32_t* out_positions, int32_t* out_count);

/* ---- HashAggregation (HashAggregationOperator; builder
 *      InMemoryHashAggregationBuilder.java:140-158) ---- */
typedef enum { TG_STEP_PARTIAL = 0, TG_STEP_FINAL = 1, TG_STEP_SINGLE = 2 } tg_agg_step;
typedef enum {
    TG_AGG_COUNT_STAR = 0,  /* CountAggregation */
    TG_AGG_COUNT_COL  = 1,
    TG_AGG_SUM_F64    = 2,  /* DoubleSumAggregation.java:37-45 */
    TG_AGG_SUM_I64    = 3,  /* BigintSumAggregation */
    TG_AGG_AVG_F64    = 4,  /* DoubleAverageAggregations.java:38-63 (state: count,sum) */
    /* exact fixed-point double sum: the caller asserts value*2^scale_pow is
     * an integer for every input (true for TPC-H money/discount columns,
     * DESIGN.md §4) and |sum*2^scale_pow| < 2^127. State is a 128-bit
     * integer accumulated with carry-propagating 64-bit atomics:
     * order-INDEPENDENT, so the result is deterministic and correctly
     * rounded — stronger than the reference's DoubleSumAggregation, equal
     * to its decimal SUM semantics. */
    TG_AGG_SUM_F64_EXACT = 5,
    /* min/max over BIGINT (MinAggregationFunction/MaxAggregationFunction for
     * BIGINT inputs): null-skipping; state = sign-bit-biased u64 extremes */
    TG_AGG_MIN_I64 = 6,
    TG_AGG_MAX_I64 = 7
} tg_agg_fn;
typedef struct tg_agg_spec {
    int32_t fn;
    int32_t input_channel;
    int32_t scale_pow;      /* TG_AGG_SUM_F64_EXACT only; else 0 */
    /* masked aggregation (AggregationMask analog, predicate form): the row
     * contributes to THIS aggregate only when col[mask_gt_a] > col[mask_gt_b]
     * (filter clause lowered into the accumulator; -1 = unmasked). Lets one
     * pass compute filtered and unfiltered aggregates side by side. */
    int32_t mask_gt_a;
    int32_t mask_gt_b;
    int32_t _pad;
} tg_agg_spec;

/* StreamingAggregationOperator analog: input grouped (clustered) by a
 * single non-null BIGINT key channel — no hash table; group ids are run
 * indexes (first-occurrence row order), page-spanning runs continue. */
tg_status tg_streaming_aggregation_create(tg_session*, int32_t key_channel,
    const tg_agg_spec* aggs, int32_t n_aggs, int32_t step, tg_operator** out);
tg_status tg_hash_aggregation_create(tg_session*,
    const int32_t* group_channels, int32_t n_group_channels,
    const int32_t* group_types,               /* tg_type per group channel */
    const tg_agg_spec* aggs, int32_t n_aggs,
    int32_t step,                              /* tg_agg_step */
    tg_operator** out);

/* ---- Hash join (HashBuilderOperator nonspilling + LookupJoinOperator) ---- */
typedef struct tg_join_bridge tg_join_bridge;   /* JoinBridgeManager analog */
tg_status tg_join_bridge_create(tg_session*, tg_join_bridge** out);
void      tg_join_bridge_close(tg_join_bridge*);

tg_status tg_hash_builder_create(tg_session*, tg_join_bridge*,
    const int32_t* build_types, int32_t n_build_channels,
    const int32_t* key_channels, int32_t n_key_channels,
    const int32_t* output_channels, int32_t n_output_channels,
    tg_operator** out);

/* join_type: 0 = inner, 1 = probe-outer (LEFT: unmatched probe rows emit
 * one row with a NULL build side — LookupJoinOperators probe-outer) */
tg_status tg_lookup_join_create_ex(tg_session*, tg_join_bridge*,
    const int32_t* probe_types, int32_t n_probe_channels,
    const int32_t* key_channels, int32_t n_key_channels,
    const int32_t* probe_output_channels, int32_t n_probe_output,
    int32_t join_type, tg_operator** out);
tg_status tg_lookup_join_create(tg_session*, tg_join_bridge*,
    const int32_t* probe_types, int32_t n_probe_channels,
    const int32_t* key_channels, int32_t n_key_channels,
    const int32_t* probe_output_channels, int32_t n_probe_output,
    tg_operator** out);

/* ---- Partitioned output (PagePartitioner.java:134-330) ----
 * Computes per-row partitions with the canonical row hash
 * (InterpretedHashGenerator.java:57-110, combine 31*h, bigint xxmix) and
 * splits the page into per-partition pages. Transport (RCCL all-to-all)
 * happens above this ABI in the rank runner. */
tg_status tg_page_partitioner_create(tg_session*,
    const int32_t* types, int32_t n_channels,
    const int32_t* partition_channels, int32_t n_partition_channels,
    int32_t partition_count,
    tg_operator** out);
/* after add_input, fetch partition p's accumulated page */
tg_status tg_page_partitioner_get_partition(tg_operator*, int32_t partition,
                                            tg_page* out_page);

/* ---- synchronous row-hash helper (tests/parity): canonical row hash ---- */
tg_status tg_hash_rows(tg_session*, const tg_page* page,
                       const int32_t* channels, int32_t n_channels,
                       uint64_t* out_hashes /* host, length position_count */);

/* ---- TPC-H device generator (bench inputs; restates io.trino.tpch v1.4
 *      column streams — plugin/trino-tpch TpchRecordSet.java call sites) ---- */
typedef struct tg_tpch_lineitem_cols {
    /* all DEVICE pointers, length row_count */
    int64_t row_count;
    int64_t* orderkey;      /* may be NULL if not requested */
    int32_t* shipdate;      /* epoch days */
    double*  quantity;
    double*  extendedprice;
    double*  discount;
    double*  tax;
    uint8_t* returnflag;    /* dictionary id: 0=A 1=N 2=R */
    uint8_t* linestatus;    /* dictionary id: 0=F 1=O */
    int32_t* commitdate;    /* optional (flags bit 1) */
    int32_t* receiptdate;   /* optional (flags bit 1) */
    int64_t* partkey;       /* optional (flags bit 2) */
    uint8_t* shipmode;      /* optional (flags bit 3): dictionary id 0..6 =
                               REG AIR,AIR,RAIL,TRUCK,MAIL,FOB,SHIP */
    int64_t* tp_cents;      /* optional (flags bit 4): this line's
                               o_totalprice contribution in cents (dbgen
                               mk_order truncation; sums to o_totalprice) */
    int64_t* suppkey;       /* optional (flags bit 5): partsupp-bridge
                               l_suppkey (canonical-row verified) */
    uint8_t* shipinstruct;  /* optional (flags bit 6): dictionary id 0..3 =
                               DELIVER IN PERSON,COLLECT COD,TAKE BACK
                               RETURN,NONE (pinned) */
} tg_tpch_lineitem_cols;

/* Generate lineitem rows for orders [order_start, order_start+order_count)
 * (1-based dense order index, part of 1,500,000×SF orders). Buffers must hold
 * the actual row count (tg_tpch_lineitem_rows); row_count is set on return. */
tg_status tg_tpch_gen_lineitem(tg_session*, double scale_factor,
    int64_t order_start, int64_t order_count,
    tg_tpch_lineitem_cols* cols /* in: buffers, out: row_count */);
tg_status tg_tpch_lineitem_rows(tg_session*, double scale_factor,
    int64_t order_start, int64_t order_count, int64_t* row_count_out,
    int64_t** dev_offsets_out /* optional; hipFree */);
/* allocate device buffers of the exact size and generate (bench/tests);
 * flags: bit0 = orderkey, bit1 = commitdate+receiptdate, bit2 = partkey */
tg_status tg_tpch_lineitem_alloc(tg_session*, double scale_factor,
    int64_t order_start, int64_t order_count, int flags,
    tg_tpch_lineitem_cols* out);
tg_status tg_tpch_lineitem_free(tg_session*, tg_tpch_lineitem_cols*);
tg_status tg_tpch_gen_orders(tg_session*, double scale_factor,
    int64_t order_start, int64_t order_count,
    int64_t* dev_orderkey, int64_t* dev_custkey, int32_t* dev_orderdate,
    uint8_t* dev_priority /* nullable; ids 0..4 = 1-URGENT..5-LOW */);
tg_status tg_tpch_gen_customer(tg_session*, double scale_factor,
    int64_t cust_start, int64_t cust_count,
    int64_t* dev_custkey, uint8_t* dev_mktsegment,
    uint8_t* dev_nationkey /* 0..24 or NULL */,
    int64_t* dev_acctbal_cents /* c_acctbal*100 or NULL */);
tg_status tg_tpch_gen_supplier(tg_session*, double scale_factor,
    int64_t supp_start, int64_t supp_count,
    int64_t* dev_suppkey, uint8_t* dev_nationkey /* 0..24 */);
tg_status tg_tpch_gen_part(tg_session*, double scale_factor,
    int64_t part_start, int64_t part_count,
    int64_t* dev_partkey, int16_t* dev_type_id /* 0..149 (SMALLINT: 150 ids
    overflow signed TINYINT); PROMO = >=125 */);

/* round 2: extended tables (streams pinned in oracle/tpch_text.h against
 * the reference's own sf0.01 dataset + SF1 answer fixtures).
 * part2: type 0..149, brand 11..55, container 0..39, name_ids = 5 color
 * ids/row (dists.dss colors, alphabetical). partsupp: 4 rows per part in
 * dbgen bridge order. orders3 adds derived o_orderstatus (0=F 1=O 2=P),
 * o_totalprice cents and o_comment pool slices. */
tg_status tg_tpch_gen_part2(tg_session*, double scale_factor,
    int64_t part_start, int64_t part_count, int64_t* d_partkey,
    int16_t* d_type, uint8_t* d_brand, int32_t* d_size, uint8_t* d_container,
    uint8_t* d_name_ids, int64_t* d_retail_cents);
tg_status tg_tpch_gen_partsupp(tg_session*, double scale_factor,
    int64_t part_start, int64_t part_count, int64_t* d_partkey,
    int64_t* d_suppkey, int32_t* d_availqty, int64_t* d_supplycost_cents);
tg_status tg_tpch_gen_supplier2(tg_session*, double scale_factor,
    int64_t supp_start, int64_t supp_count, int64_t* d_suppkey,
    uint8_t* d_nationkey, int64_t* d_acctbal_cents);
tg_status tg_tpch_gen_orders3(tg_session*, double scale_factor,
    int64_t order_start, int64_t order_count, int64_t* d_orderkey,
    int64_t* d_custkey, int32_t* d_orderdate, uint8_t* d_priority,
    uint8_t* d_orderstatus, int64_t* d_totalprice_cents,
    int64_t* d_cmnt_off, int32_t* d_cmnt_len);
/* 300 MiB dbgen text pool, device-resident (built host-side, cached) */
tg_status tg_tpch_pool(tg_session*, const uint8_t** d_pool);
/* LIKE '%a%b%' (no '_') over pool slices / var-width columns -> 0/1 flags */
tg_status tg_pool_like_flags(tg_session*, const int64_t* d_offs,
    const int32_t* d_lens, int64_t n, const char* pattern, uint8_t* d_flags);
tg_status tg_varchar_like_flags(tg_session*, const uint8_t* d_bytes,
    const int32_t* d_offsets, int64_t n, const char* pattern, uint8_t* d_flags);
/* s_comment var-width column incl. the BBB "Customer ...Complaints/
 * Recommends" overlay (10 rows per 10,000 suppliers) */
tg_status tg_tpch_gen_supplier_comments(tg_session*, double scale_factor,
    int64_t supp_start, int64_t supp_count, int32_t** d_offsets_out,
    uint8_t** d_bytes_out);
/* host-side string materialization for final output assembly (few rows) */
tg_status tg_tpch_supplier_strings(double sf, const int64_t* keys, int32_t n,
    int32_t stride, char* name, char* address, char* phone, char* comment,
    int64_t* acctbal_cents, int32_t* nationkey);
tg_status tg_tpch_customer_strings(double sf, const int64_t* keys, int32_t n,
    int32_t stride, char* name, char* address, char* phone, char* comment,
    int64_t* acctbal_cents, int32_t* nationkey);
tg_status tg_tpch_part_strings(double sf, const int64_t* keys, int32_t n,
    int32_t stride, char* name, char* mfgr, char* brand, char* type,
    char* container);
tg_status tg_tpch_nation_name(int32_t nationkey, char out[32]);
int32_t tg_tpch_nation_region(int32_t nationkey);
/* p_name predicate flags over the 5 color ids per part (green=33 forest=28) */
tg_status tg_tpch_part_name_flag(tg_session*, const uint8_t* d_name_ids,
    int64_t n, int32_t color_id, int32_t first_only, uint8_t* d_flags);
/* fused dynamic filter (sql/gen/columnar/DynamicPageFilter.java +
 * operator/DynamicFilterSourceOperator.java analog): request a dense-range
 * membership bitmap over the build keys BEFORE the build finishes, then
 * create the probe-side scan with the bridge attached — the membership test
 * runs inside the scan's filter kernel after the static predicate. Best
 * effort like the reference: generic/multi-channel keys or ranges beyond
 * 2^33 skip the bitmap and the scan runs the static filter alone. */
tg_status tg_join_bridge_request_bitmap(tg_join_bridge*);
tg_status tg_filter_project_create_df(tg_session*, const tg_expr* filter,
    const tg_expr* projections, const int32_t* proj_out_types, int32_t n_proj,
    tg_join_bridge* df_bridge, int32_t df_key_channel, tg_operator** out);

/* ---- native Parquet reader (csrc/parquet.cpp; mirrors
 * lib/trino-parquet/reader/ParquetReader.java surface) ---- */
typedef struct tg_parquet_file tg_parquet_file;
tg_status tg_parquet_open(tg_session*, const char* path, tg_parquet_file**);
void tg_parquet_close(tg_parquet_file*);
int64_t tg_parquet_num_rows(tg_parquet_file*);
int32_t tg_parquet_num_columns(tg_parquet_file*);
const char* tg_parquet_column_name(tg_parquet_file*, int32_t);
int32_t tg_parquet_physical_type(tg_parquet_file*, int32_t);
tg_status tg_parquet_read_column(tg_session*, tg_parquet_file*, int32_t col,
    void* out_values, uint64_t* out_valid, int32_t* out_ids,
    uint8_t* out_dict_bytes, int64_t dict_bytes_cap,
    int32_t* out_dict_offsets, int32_t* out_dict_count);
/* multi-column parallel decode (columns x row groups nested) */
tg_status tg_parquet_read_columns(tg_session*, tg_parquet_file*,
    const int32_t* cols, int32_t n_cols, void** out_values,
    uint64_t** out_valid, int32_t** out_ids, uint8_t** out_dict_bytes,
    const int64_t* dict_caps, int32_t** out_dict_offsets,
    int32_t** out_dict_counts);

tg_status tg_copy_dtod(tg_session*, void* dst_dev, const void* src_dev,
                       int64_t bytes);

/* adaptive partial aggregation (operator/aggregation/partial/
 * PartialAggregationController.java:34-100 analog): one controller is
 * shared across the PARTIAL-step hash aggregations of a plan node. After
 * >= 1.5x max_partial_bytes of input has been sampled, if the ratio of
 * unique output rows to input rows exceeds the threshold (reference
 * session default 0.8 — adaptive_partial_aggregation_unique_rows_ratio_
 * threshold), partial aggregation flips to pass-through: pages are
 * re-shaped into the partial-state channel layout with no hash-table work
 * and the FINAL stage does the grouping. Re-enabled after 200x more bytes
 * (same constants as the reference). on_flush is exposed so host drivers
 * (and CPU tests) can feed flush statistics directly. Restrictions:
 * fixed-width group keys, unmasked aggregates. */
typedef struct tg_pa_controller tg_pa_controller;
tg_status tg_pa_controller_create(int64_t max_partial_bytes,
    double unique_rows_ratio_threshold, tg_pa_controller** out);
void tg_pa_controller_close(tg_pa_controller*);
int32_t tg_pa_controller_disabled(tg_pa_controller*);
tg_status tg_pa_controller_on_flush(tg_pa_controller*, int64_t bytes,
    int64_t rows, int64_t unique_rows, int32_t have_unique);
tg_status tg_hash_aggregation_set_controller(tg_operator*, tg_pa_controller*);

/* MarkDistinctOperator analog: appends a BOOLEAN channel marking each
 * row's first occurrence over the key channels (streaming pass-through) */
tg_status tg_mark_distinct_create(tg_session*, const int32_t* key_channels,
    int32_t n_key_channels, const int32_t* key_types, tg_operator**);

/* dense-range single-BIGINT-key aggregation (direct array state, one
 * atomic per row — or an atomic pair for the 128-bit exact sum; groups
 * emit in key order). For COUNT/SUM_I64/SUM_F64_EXACT over keys with
 * known dense statistics (generated custkeys/suppkeys). */
tg_status tg_dense_aggregation_create(tg_session*, int32_t key_channel,
    int64_t key_min, int64_t key_max, const tg_agg_spec* agg, tg_operator**);

/* sort-based DISTINCT of a non-negative BIGINT device column (radix sort +
 * unique compaction; `bits` = key width to sort). For near-all-unique
 * dedups (count(DISTINCT ...)) where hash aggregation pays random keystore
 * probes per row. d_out sized n; *out_n gets the distinct count. */
tg_status tg_dedup_i64(tg_session*, const int64_t* d_in, int64_t n,
    int32_t bits, int64_t* d_out, int64_t* out_n);

/* stream timer (HIP events on the session stream) for bench rooflines */
tg_status tg_timer_start(tg_session*);
tg_status tg_timer_stop(tg_session*, double* elapsed_ms);

/* coarse device-memory accounting (LocalMemoryContext analog): bytes ever
 * pooled and bytes currently cached; live = total - cached */
tg_status tg_session_memory(tg_session*, int64_t* total_bytes, int64_t* cached_bytes);

/* device buffer management for host pipeline drivers (pool-backed) */
tg_status tg_device_malloc(tg_session*, void** out, int64_t bytes);
tg_status tg_device_free(tg_session*, void* p);

/* ---- TopN (operator/TopNOperator.java): ORDER BY ... LIMIT ----
 * sort_desc[i]: 0 = ASC (nulls last), 1 = DESC (nulls first) */
tg_status tg_topn_create(tg_session*,
    const int32_t* types, int32_t n_channels,
    const int32_t* sort_channels, const int32_t* sort_desc, int32_t n_sort,
    int32_t limit, tg_operator** out);

/* ---- semi join (operator/HashSemiJoinOperator.java): appends a BOOLEAN
 * matched channel to the probe page (NULL for null probe keys) ---- */
/* SetBuilderOperator analog (ChannelSet): semi-join membership source.
 * Dense single-BIGINT key ranges build a bitmap; sparse fall back to the
 * positional index. The bridge then serves tg_semi_join_create only. */
tg_status tg_set_builder_create(tg_session*, tg_join_bridge*,
    const int32_t* build_types, int32_t n_build_channels,
    int32_t key_channel, tg_operator** out);
tg_status tg_semi_join_create(tg_session*, tg_join_bridge*,
    int32_t key_channel, tg_operator** out);

/* ---- dynamic filter source (DynamicFilterSourceOperator /
 * sql/gen/columnar/DynamicPageFilter.java analog): min/max over the build
 * side's non-null keys, valid after the builder's finish ---- */
tg_status tg_join_bridge_key_range(tg_join_bridge*, int64_t* key_min,
                                   int64_t* key_max, int64_t* key_rows);

/* test helpers */
tg_status tg_copy_dtoh(tg_session*, void* dst, const void* src, int64_t bytes);
tg_status tg_copy_htod(tg_session*, void* dst_dev, const void* src_host, int64_t bytes);

/* ---- fused TPC-H Q1 pipeline (the north-star benchmark kernel):
 * scan+filter(shipdate<=cutoff)+group(returnflag,linestatus)+7 aggregates,
 * one pass over HBM (38 B/row algorithmic). Results per (rf,ls) combo. ---- */
typedef struct tg_q1_result {
    /* indexed [rf*2+ls], rf in {A,N,R}=0,1,2; ls in {F,O}=0,1 */
    double sum_qty[6], sum_base[6], sum_disc_price[6], sum_charge[6];
    double avg_qty[6], avg_price[6], avg_disc[6];
    double sum_disc[6];
    int64_t count[6];
    double elapsed_ms;          /* kernel time, HIP events */
    /* raw integer accumulators (per combo: base,dp,ch,disc as u128 lo/hi
     * pairs then qty,cnt), for exact cross-GPU merging */
    uint64_t raw[60];
} tg_q1_result;

tg_status tg_q1_run(tg_session*, const tg_tpch_lineitem_cols* cols,
                    int32_t shipdate_cutoff, tg_q1_result* out);
/* parity mode: reference sequential accumulation order (page-sized inputs) */
tg_status tg_q1_run_naive(tg_session*, const tg_tpch_lineitem_cols* cols,
                          int32_t shipdate_cutoff, tg_q1_result* out);

#ifdef __cplusplus
}
#endif
#endif /* TRINO_GPU_H */
