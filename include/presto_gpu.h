/* presto_gpu.h — public C-ABI of the MI355X-native Presto hot-path library.
 *
 * This is the drop-in boundary of SURVEY.md §8b: a handle-based operator
 * lifecycle mirroring the reference's Operator interface
 * (presto-main-base/.../operator/Operator.java:20-102, constructed via
 * OperatorFactory.createOperator and driven by Driver.processInternal:402),
 * with Page data crossing the seam as flat column descriptors matching the
 * reference's Block layouts (presto-common/.../block/LongArrayBlock.java:38-52,
 * IntArrayBlock, ByteArrayBlock, DictionaryBlock.java:53-64).
 *
 * In production the Java side binds this ABI through a thin JNI veneer
 * (see INTEGRATION.md for the binding a presto-main maintainer would add);
 * in this repo the same ABI is driven by the Python harness in presto_amd/
 * and by tests/ replicating OperatorAssertion.toPages
 * (presto-main-base/src/test/.../OperatorAssertion.java:62-176).
 *
 * Contract (verified by the reference's Driver, restated here):
 *  - one handle == one logical driver: all calls on a handle are
 *    single-threaded (Driver.processInternal holds an exclusive lock);
 *  - pg_op_add_input only when pg_op_needs_input returns 1
 *    (Driver.java:446-458); input pages are borrowed for the call;
 *  - pg_op_get_output may produce nothing (*out == NULL); output pages are
 *    owned by the library until the next pg_op_get_output/pg_op_destroy;
 *  - pg_op_finish is idempotent; cleanup via pg_op_destroy (close()).
 *  - build->probe bridging uses table handles (mirrors
 *    PartitionedLookupSourceFactory.java:149,181).
 *
 * Errors: negative return, message via pg_last_error().
 * All compute requires an AMD gfx950 GPU; calls fail loudly without one
 * (there is no CPU fallback in this library).
 */
#ifndef PRESTO_GPU_H
#define PRESTO_GPU_H
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define PG_ABI_VERSION 1

/* ---- status ---- */
typedef int32_t pg_status; /* 0 ok, <0 error */
#define PG_OK 0
#define PG_ERR (-1)
const char* pg_last_error(void);

/* ---- device helpers (memory plumbing for the harness; tensors may also be
 * allocated by the caller, e.g. torch, and passed as raw device pointers) */
pg_status pg_device_count(int32_t* out);
pg_status pg_device_sync(void);
pg_status pg_device_malloc(int64_t bytes, void** out);
pg_status pg_device_free(void* p);
pg_status pg_memcpy_h2d(void* dst, const void* src, int64_t bytes);
pg_status pg_memcpy_d2h(void* dst, const void* src, int64_t bytes);
pg_status pg_memcpy_d2d(void* dst, const void* src, int64_t bytes);
/* HIP-event elapsed ms of the most recent hot-path kernel launch (fused
 * scan+filter+aggregate, or probe+aggregate) — measurement plumbing for
 * bench.py's roofline leg */
double pg_last_hot_kernel_ms(void);
/* max hot-region ms since pg_hot_reset (for pipelines with several hot
 * launches per step: the max is the dominant one) */
double pg_hot_max_ms(void);
void pg_hot_reset(void);

/* ---- Page / Block descriptors (SURVEY.md §8b struct) ----
 * Concrete layouts follow the Java blocks: fixed-width values array +
 * optional byte-per-position null mask (LongArrayBlock.java:38-52). */
typedef enum {
    PG_T_U8 = 0,  /* ByteArrayBlock / dictionary codes */
    PG_T_I32 = 1, /* IntArrayBlock / DateType days */
    PG_T_I64 = 2, /* LongArrayBlock / BigintType */
    PG_T_F64 = 3, /* LongArrayBlock bits / DoubleType */
    PG_T_I128 = 5, /* Int128ArrayBlock (Int128ArrayBlock.java): 16-byte
                       little-endian (low, high) pairs — decimal(p>18,s)
                       unscaled values.  Carried through staging, IDENT
                       projection/emit and the INT128_ARRAY wire encoding;
                       decimal arithmetic accumulates through the exact
                       int128 totals of the aggregation paths
                       (fixed128.h, UnscaledDecimal128Arithmetic.java:770
                       add semantics). */
    PG_T_VARBIN = 4, /* VariableWidthBlock (VariableWidthBlock.java:48-61):
                        data = bytes, offsets = int32[n_rows+1] (element i
                        spans bytes [offsets[i], offsets[i+1])).  Supports
                        EQ/NE/CONTAINS/PREFIX predicates, hashing/
                        partitioning (XxHash64 per
                        AbstractVariableWidthBlock.java:102-105) and
                        IDENT projection/emit (two-pass gather). */
} pg_type;

typedef struct {
    int32_t tag;            /* pg_type */
    int32_t on_device;      /* 1: data is a device pointer */
    void* data;             /* values array (VARBIN: the byte buffer) */
    const uint8_t* null_mask; /* optional, 1 byte/pos, 1 = null; may be NULL */
    const int32_t* offsets; /* VARBIN only: offsets (n_rows+1, or dict_n+1
                               for dictionary columns) */
    /* dictionary encoding (DictionaryBlock.java:60-86): when dict_ids is
     * non-NULL the column is a dictionary VARBIN block — data/offsets
     * describe the dict_n dictionary entries and dict_ids[n_rows] maps
     * each position to an entry.  Predicates, hashing/partitioning and
     * emit read through the ids. */
    const int32_t* dict_ids;
    int32_t dict_n;
} pg_col;

typedef struct {
    int64_t n_rows;
    int32_t n_cols;
    pg_col cols[32];
} pg_page;

/* ---- plan blobs ----
 * The reference JIT-compiles per-query filter/projection/accumulator classes
 * at runtime (sql/gen/PageFunctionCompiler.java:126, AccumulatorCompiler,
 * JoinCompiler).  The MI355X-native analog is a set of ahead-of-time
 * specialized HIP kernels selected by these plan descriptors. */

typedef enum { PG_CMP_LT = 0, PG_CMP_LE, PG_CMP_GT, PG_CMP_GE, PG_CMP_EQ,
               PG_CMP_NE,
               /* VARBIN only — the LIKE '%w%' / 'w%' pushdowns
                * (LikeFunctions.java:64-77 likeVarchar for patterns
                * without '_' reduce to substring/prefix search) */
               PG_CMP_CONTAINS, PG_CMP_PREFIX,
               /* ordered two-substring LIKE '%a%b%' (and its negation —
                * Q13's o_comment NOT LIKE '%special%requests%'):
                * sval holds a then b concatenated, slen = len(a),
                * ival = len(b) */
               PG_CMP_CONTAINS2, PG_CMP_NOT_CONTAINS2 } pg_cmp;

typedef struct {
    int32_t col;   /* input channel */
    int32_t op;    /* pg_cmp */
    int64_t ival;  /* compare value for integer columns */
    double dval;   /* compare value for f64 columns */
    char sval[40]; /* VARBIN: compare bytes (EQ/NE/CONTAINS/PREFIX;
                      CONTAINS2 packs both patterns) */
    int32_t slen;
    int32_t rhs_col; /* 0: compare against the constant; k>0: compare
                        against integer channel k-1 PLUS the constant
                        (col OP col + ival / dval — e.g. Q5's
                        c_nationkey = s_nationkey, TPC-DS Q72's
                        d3.d_date > d1.d_date + 5).  Zero-initialized
                        plans keep constant semantics. */
} pg_pred;

/* projection expressions (PageProjection analogs) */
typedef enum {
    PG_PROJ_IDENT = 0,        /* column a */
    PG_PROJ_DISC_PRICE = 1,   /* a * (1 - b)            (Q1/Q3 revenue) */
    PG_PROJ_CHARGE = 2,       /* a * (1 - b) * (1 + c)  (Q1 charge) */
    PG_PROJ_MUL = 3,          /* a * b                  (Q6 revenue) */
    PG_PROJ_DIV = 4,          /* a / b (f64 emit only — e.g. a grouped
                                 sum/count mean materialized for a
                                 downstream compare, Q21's only-late
                                 supplier) */
    PG_PROJ_KEYSHL = 5,       /* (a << c) | b — composite grouping keys
                                 (c = shift constant, i64 emit; the
                                 codegen analog of CombineHashFunction
                                 key packing for multi-channel group-bys,
                                 e.g. Q16's (brand,type,size,suppkey)) */
    PG_PROJ_SHR = 6,          /* a >> c — composite key extraction */
    PG_PROJ_SUBDIV = 7,       /* (a - b) / c with b, c CONSTANTS (i64
                                 emit; floor division of non-negative
                                 differences) — planner expressions like
                                 date -> week_seq ((date - day0) / 7) or
                                 multiplicity - 1 */
    PG_PROJ_KEYSHL_DIV = 8,   /* (a << shift) | (b / div) with
                                 c = (shift << 16) | div — composite keys
                                 over a derived dimension (e.g. TPC-DS
                                 Q72's (item, week) key in ONE pass:
                                 item << 14 | date/7) */
} pg_proj_kind;

typedef struct {
    int32_t kind; /* pg_proj_kind */
    int32_t a, b, c; /* input channels */
} pg_proj;

/* aggregate functions over a projection */
typedef enum {
    PG_AGG_COUNT = 0,    /* CountAggregation.java:34 */
    PG_AGG_SUM_F64 = 1,  /* DoubleSumAggregation (deterministic schedule,
                            DESIGN.md §determinism) */
    PG_AGG_SUM_DEC = 2,  /* exact decimal ticks; scale from dec_scale
                            (hive-decimal semantics of the golden vectors) */
    PG_AGG_SUM_I64 = 3,  /* LongSumAggregation.java:33-37 */
    PG_AGG_MIN = 4,      /* MinAggregationFunction: decimal-ticks/i64 min
                            in decimal mode, f64 min in f64 mode */
    PG_AGG_MAX = 5,
} pg_agg_func;

typedef struct {
    int32_t func;     /* pg_agg_func */
    pg_proj proj;     /* input expression */
    int32_t dec_scale; /* for SUM_DEC: ticks = round(value * 10^dec_scale) */
} pg_agg;

/* -------- operator kinds + plans -------- */
typedef enum {
    PG_OP_FILTER_PROJECT = 1, /* ScanFilterAndProjectOperator.java:67 +
                                 PageProcessor.java:112,299-343 */
    PG_OP_HASH_AGG_SMALL = 2, /* HashAggregationOperator.java:56 with
                                 low-cardinality dict-u8 keys (Q1 shape) */
    PG_OP_HASH_BUILD = 3,     /* HashBuilderOperator.java:55 */
    PG_OP_LOOKUP_JOIN = 4,    /* LookupJoinOperator.java:481-604 (optionally
                                 fused with grouped SUM into the table —
                                 Q3's join+partial-agg pipeline) */
    PG_OP_TOPN = 5,           /* TopNOperator.java:32,90-111 */
    PG_OP_PARTITION = 6,      /* PartitionedOutputOperator.partitionPage:394 /
                                 LocalExchange partition split */
    PG_OP_GROUPBY_MULTI = 7,  /* MultiChannelGroupByHash.java:300-380 +
                                 InMemoryHashAggregationBuilder: general
                                 grouped aggregation over 1..4 key
                                 channels of mixed type at arbitrary
                                 cardinality (see pg_plan_groupby) */
} pg_op_kind;

#define PG_MAX_PRED 8
#define PG_MAX_AGG 8
#define PG_MAX_KEYVALS 8

typedef struct {
    int32_t n_preds;
    pg_pred preds[PG_MAX_PRED]; /* conjunction */
    int32_t n_proj;
    pg_proj proj[16];
    int64_t semijoin_table; /* >0: keep only rows whose semijoin_col key is
                               in that key-set table (EXISTS pushdown, e.g.
                               Q4); 0 = unused */
    int32_t semijoin_col;
    int32_t semijoin_anti;  /* 1: keep rows whose key is NOT in the set
                               (NOT-EXISTS pushdown, e.g. Q22's customers
                               without orders — LookupJoinOperator's
                               probe-side anti join) */
} pg_plan_filter_project;

typedef struct {
    /* up to two u8 key channels with enumerated code values; group id =
     * idx(key0)*n_vals1 + idx(key1); results emitted in that order,
     * restricted to non-empty groups */
    int32_t n_preds;
    pg_pred preds[PG_MAX_PRED]; /* fused pre-filter (scan+filter+agg) */
    int32_t n_keys;             /* 0 (single global group), 1 or 2 */
    int32_t key_col[2];
    int32_t n_vals[2];
    uint8_t key_vals[2][PG_MAX_KEYVALS];
    int32_t n_aggs;
    pg_agg aggs[PG_MAX_AGG];
    int32_t drop_unlisted_keys; /* 1: rows whose key is outside key_vals are
                                   dropped (dictionary-filter pushdown, e.g.
                                   Q5's region membership); 0: error */
} pg_plan_hash_agg_small;

typedef struct {
    int32_t n_preds;
    pg_pred preds[PG_MAX_PRED]; /* fused pre-filter on build input */
    int32_t key_col;            /* bigint key channel */
    int64_t semijoin_table;     /* >=0: keep only rows whose column
                                   semijoin_col matches that table's key set
                                   (Q3: orders ⋉ building-customers) */
    int32_t semijoin_col;
    int32_t n_payload;
    int32_t payload_col[4];     /* i32/i64 payload channels stored per row */
    int64_t capacity_hint;      /* expected distinct build rows */
    int32_t key_set_only;       /* 1: build a key SET (no payload slots) */
    int32_t dense_array;        /* 1: keys are dense 1..capacity_hint —
                                   store the single u8 payload in a direct
                                   array indexed by key-1 (no hashing);
                                   e.g. suppkey -> s_nationkey */
    /* agg_table only: fetch the u8 payload THROUGH another agg table
     * (star-schema dimension join fused into the build: e.g. Q5's orders
     * build stores customer nation = cust_table[o_custkey].payload; rows
     * whose lookup key misses are dropped).  0 = unused. */
    int64_t payload_lookup_table;
    int32_t payload_lookup_key_col;
    int32_t agg_table;          /* 1: table feeds a fused-agg probe only
                                   (LOOKUP_JOIN mode 1).  Rows are inserted
                                   directly during addInput (single scan,
                                   payloads stored per slot, keys must be
                                   unique); join-emit probing (chains) is
                                   rejected.  capacity_hint must be >= the
                                   number of inserted rows (errors out
                                   otherwise). */
    /* agg_table only: store the single payload in the low pack_bits of
     * the key slot word (slot = key << pack_bits | payload) so one CAS
     * carries key AND payload — one random line per insert and per probe
     * hit instead of two.  Caller guarantees 0 <= key < 2^(63-pack_bits)
     * and 0 <= payload < 2^pack_bits (checked; insert errors out
     * otherwise).  The analog of the reference's SyntheticAddress
     * packing (SyntheticAddress.java:23-36), applied to slot payloads.
     * 0 = unpacked.  Requires n_payload == 1. */
    int32_t pack_bits;
    /* agg_table only: capacity multiplier x10 (0 = default 20, i.e.
     * cap = next_pow2(2 x capacity_hint), fill <= ~0.5 — sized for
     * short linear-probe clusters on miss-heavy probes).  Probes that
     * ALWAYS hit (e.g. lineitem -> its order) can size tighter: 13
     * gives fill <= ~0.77 and halves the table/accumulator footprint. */
    int32_t fill_x10;
    /* > 0: also build a key-presence BITMAP over [1, bitmap_max_key]
     * (one bit per possible key; the analog of the reference's bigint
     * dynamic-filter / bloom pre-filter in front of a join probe,
     * DynamicFilterSourceOperator.java:214-258 collecting and
     * LookupJoinOperator applying it).  Probes test the bit BEFORE
     * computing the bucket hash: on miss-heavy probes of primary-key
     * domains (orderkey spans 4 x orders rows, so the SF100 bitmap is
     * 75 MB — L3-resident) this replaces the hash + tag-line dependent
     * chain with one cached load for ~90% of rows.  Build keys outside
     * [1, bitmap_max_key] are an error (the bitmap cannot represent
     * them, so probes would wrongly reject).  0 = no bitmap. */
    int64_t bitmap_max_key;
    /* 1: a RANGE-GROUP table — the group-by domain is the dense key
     * range [1, capacity_hint] itself (the perfect-hash GroupByHash
     * analog for primary-key-shaped bigint group channels, cf.
     * BigintGroupByHash.java:66-117 whose probe is exactly a
     * value->bucket map).  No build input is scanned and no key slots
     * are stored: a mode-1 fused-agg probe indexes its accumulators by
     * key-1 directly (keys outside the range are misses, matching
     * inner-join-with-FK semantics), and group extraction reconstructs
     * the key from the slot index.  For probe inputs CLUSTERED by the
     * key (lineitem by orderkey) the accumulator atomics become
     * near-sequential instead of hash-scattered. */
    int32_t range_group;
    /* dense_array only: constant added to the payload value at fill
     * time (probe-side consumers see value+bias).  Lets 0-valued
     * payloads (nationkey 0, priority 0) coexist with "presence =
     * nonzero" dense probing; the pipeline accounts for the bias in
     * its downstream predicates. */
    int32_t dense_payload_bias;
} pg_plan_hash_build;

typedef struct {
    int64_t table;    /* handle from a finished PG_OP_HASH_BUILD */
    int32_t n_preds;
    pg_pred preds[PG_MAX_PRED]; /* fused pre-filter on probe input */
    int32_t key_col;
    /* mode 2: fully fused probe + dense lookup + equality + small-key
     *   grouped SUM (the specialization for Q5's local-supplier shape):
     *   for each probe row passing preds: g1 = table[key_col] payload
     *   (u8), g2 = dense table2[table2_key_col], keep when g1 == g2 and
     *   g2 is one of group_vals; acc[g2] += proj (exact ticks + fx128).
     *   get_output after finish: host page [group u8, sum_dec i64 ticks,
     *   sum_f64, count i64] in group_vals order (non-empty groups).
     * mode 0: emit matched rows — output page =
     *   probe columns emit_probe_cols[] + build payloads (join emit order:
     *   probe rows ascending; within a probe row, chain head first —
     *   LookupJoinPageBuilder.appendRow:75 + ArrayPositionLinks order)
     * mode 3: fused star-join grouped SUM — probe table (chained, first
     *   payload = an i64 grouping key, e.g. orders keyed by orderkey
     *   with o_custkey payload), accumulate proj into table2's slots
     *   keyed by that payload (Q10: revenue per customer in ONE pass
     *   over lineitem).  get_output emits table2's groups page.
     * mode 1: fused grouped SUM into the build table (group = join key):
     *   for each match, table.acc += proj(probe row) in exact decimal ticks
     *   AND exact-f64 fixed-point (fixed128.h); get_output after finish
     *   emits the groups page: key i64, payloads..., sum_dec i64,
     *   sum_f64 f64 (rows = groups with >=1 match) */
    int32_t mode;
    int32_t n_emit;
    int32_t emit_probe_cols[8];
    pg_proj proj;
    int32_t dec_scale;
    /* mode 2 only: */
    int64_t table2;          /* dense_array build */
    int32_t table2_key_col;
    int32_t n_group_vals;
    uint8_t group_vals[8];
    /* mode 1 only: skip the exact-f64 (fx128) accumulator legs when the
     * consumer reads only the decimal sum + count (integer aggregates
     * such as Q17/Q18/Q21 quantity sums — halves the atomic traffic of
     * all-match probes) */
    int32_t dec_only;
    /* mode 1 only: the decimal accumulator keeps the MIN of the
     * projected ticks instead of the sum (MinAggregationFunction over a
     * grouped probe — Q2's per-part minimum supplycost).  Implies
     * dec_only. */
    int32_t dec_min;
    /* mode 1 only: MULTI-ACCUMULATOR probe — the full
     * InMemoryHashAggregationBuilder.processPage:204 analog (one
     * getGroupIds probe, then EVERY aggregator's addInput), optionally
     * with per-aggregate FILTER clauses (AggregationNode's
     * aggregation masks).  n_aggs > 0 replaces the single proj/dec legs:
     * per-slot accumulators are n_aggs int64 tick sums plus the match
     * count; agg_filter[a] >= 0 indexes a predicate in preds[] that
     * gates aggregate a alone (row-level filtering still uses
     * preds[0..n_preds)).  Overflow-checked (Math.addExact).  Output
     * page after finish: key, payloads..., agg0..agg{n-1}, count. */
    int32_t n_aggs; /* 0 = legacy single proj; 1..6 multi */
    pg_agg aggs[6];
    int32_t agg_filter[6];
    /* Multi-agg accumulator PACKING (the slot-word analog of pack_bits,
     * applied to the accumulator state): aggregates whose per-GROUP
     * totals the caller can bound (integrity facts like "<= 7 lineitems
     * per order" and "suppkey < supplier count") share ONE u64 word as
     * bit fields, so a probe flush issues one atomicAdd on one line
     * instead of one per aggregate — the multi-agg probe is atomic-op-
     * rate bound, not bandwidth bound.  acc_pack = 1 enables it;
     * acc_pack_shift[a] is the field's bit offset in word 0 or -1 for a
     * full u64 word of its own (appended after word 0 in agg order);
     * acc_pack_width[a] the field width.  The count field lives at
     * acc_pack_cnt_shift/width in word 0.  CONTRACT: the caller
     * guarantees every per-group TOTAL fits its declared field; each
     * probe flush additionally checks its own contribution against the
     * width and raises the overflow error on violation (gross width
     * mistakes fail loudly; the bound itself is the caller's integrity
     * fact, as with pack_bits).  Output schema is unchanged — group
     * extraction unpacks the fields. */
    int32_t acc_pack;
    int32_t acc_pack_shift[6];
    int32_t acc_pack_width[6];
    int32_t acc_pack_cnt_shift;
    int32_t acc_pack_cnt_width;
} pg_plan_lookup_join;

typedef struct {
    /* General multi-channel grouped aggregation — the
     * MultiChannelGroupByHash.java:300-380 analog: arbitrary group
     * cardinality over 1..4 key channels of mixed I64/I32/U8 or
     * dictionary-VARBIN type (dictionary keys group by their dictionary
     * id — DictionaryBlock identity, emitted back as I32 ids into the
     * same dictionary; requires distinct dictionary entries, i.e. a
     * proper dictionary).  Open-address table, linear probe, group hash
     * = murmur3-finalized CombineHashFunction fold of the per-channel
     * bigint hashes (CombineHashFunction.java:28-30,
     * AbstractLongType.java:137-140); multi-word keys are claimed with
     * a per-slot state word (CAS-claim, release-publish), so concurrent
     * inserts are exact — never fingerprint-approximate.
     * Aggregates: SUM_DEC/SUM_I64 (overflow-checked ticks), COUNT,
     * SUM_F64 (exact 64.64 fixed point), MIN/MAX (integer ticks), each
     * optionally gated by a FILTER predicate (agg_filter indexes
     * preds[]; row-level filtering uses preds[0..n_preds)).
     * Output after finish: key channels (input types), then per
     * aggregate its value column (I64 ticks or F64), then the group
     * row count; groups emitted slot-ascending (order is not part of
     * the contract, as with the reference's parallel drivers).
     * capacity_hint must be >= the number of distinct groups (errors
     * out past fill 0.85, like agg tables). */
    int32_t n_preds;
    pg_pred preds[PG_MAX_PRED];
    int32_t n_keys;
    int32_t key_col[4];
    int64_t capacity_hint;
    int32_t n_aggs;
    pg_agg aggs[6];
    int32_t agg_filter[6];
} pg_plan_groupby;

typedef struct {
    /* ORDER BY value DESC, date ASC, key ASC LIMIT n  (Q3 shape) */
    int32_t limit;
    int32_t val_col;  /* i64 (decimal ticks) or f64 */
    int32_t date_col; /* i32 */
    int32_t key_col;  /* i64 */
} pg_plan_topn;

typedef struct {
    int32_t n_partitions;
    int32_t key_col; /* bigint; rawHash = CombineHashFunction fold from
                        INITIAL_HASH_VALUE=0 of AbstractLongType.hash
                        (PlannerUtils.java:113, AbstractLongType.java:137-140);
                        partition = HashGenerator.java:22-29 */
    int32_t n_emit;
    int32_t emit_cols[8];
} pg_plan_partition;

/* ---- operator lifecycle (Operator.java:20-102 analog) ---- */
typedef int64_t pg_op;

pg_status pg_op_create(int32_t kind, const void* plan, int64_t plan_bytes,
                       pg_op* out);
/* needsInput() */
int32_t pg_op_needs_input(pg_op op);
/* addInput(Page) — page borrowed for the call */
pg_status pg_op_add_input(pg_op op, const pg_page* page);
/* getOutput() — *out set to an internal page (device pointers) or NULL */
pg_status pg_op_get_output(pg_op op, const pg_page** out);
/* finish() */
pg_status pg_op_finish(pg_op op);
/* isFinished() */
int32_t pg_op_is_finished(pg_op op);
pg_status pg_op_destroy(pg_op op);

/* table handle of a finished HASH_BUILD op
 * (lendPartitionLookupSource analog, HashBuilderOperator.java:534) */
pg_status pg_op_table(pg_op op, int64_t* out_table);
/* per-partition row counts after a PARTITION op's get_output */
pg_status pg_op_partition_counts(pg_op op, int64_t* counts, int32_t n);

/* destroy a table explicitly (tables outlive their build op until freed) */
pg_status pg_table_destroy(int64_t table);
/* zero an agg table's accumulators so the table (keys + chains) can be
 * reused by another fused-agg probe — the analog of reusing a lookup
 * source across probe factories (HashBuilderOperator.java:534
 * lendPartitionLookupSource is multi-consumer) */
pg_status pg_table_reset_acc(int64_t table);

/* ---- SerializedPage wire interop ----
 * Presto's exchange wire format, restated from
 * spi/page/PagesSerdeUtil.java:64-88 (metadata: positionCount i32, codec
 * marker u8, uncompressedSize i32, size i32, checksum i64, then the page
 * bytes; little-endian), checksum per computeSerializedPageChecksum:109-120
 * (CRC32 over data + marker + positionCount + uncompressedSize bytes);
 * raw page bytes per writeRawPage:45-51 (block count i32 then per block a
 * length-prefixed encoding name, BlockEncodingManager.java:96-99) with
 * encodings LONG_ARRAY / INT_ARRAY / BYTE_ARRAY
 * (common/block/LongArrayBlockEncoding.java:26-48 etc.; nulls-as-bits per
 * EncoderUtil.java:31-63; non-null values only).
 * v1 scope: uncompressed, unencrypted (codec marker 0); fixed-width
 * blocks (I64/F64 -> LONG_ARRAY, I32 -> INT_ARRAY, U8 -> BYTE_ARRAY).
 * Host-side (the node-boundary seam stays on the host). */
pg_status pg_page_serialize(const pg_page* page /* host cols */,
                            void* out, int64_t cap, int64_t* out_len);
/* round-2 wire scope: compress=1 runs the body through the LZ4 block
 * codec and sets PageCodecMarker.COMPRESSED when the ratio clears
 * PagesSerde.java:41's MINIMUM_COMPRESSION_RATIO (0.9); encodings now
 * cover LONG/INT/BYTE/INT128_ARRAY, VARIABLE_WIDTH, DICTIONARY
 * (varbin dictionaries travel as dictionaries, fixed-width ones expand
 * on read) and RLE (expanded on read). */
/* ABI drift guard: fills out[0..n) with sizeof each public struct in
 * the order {pg_col, pg_page, pg_pred, pg_proj, pg_agg,
 * pg_plan_filter_project, pg_plan_hash_agg_small, pg_plan_hash_build,
 * pg_plan_lookup_join, pg_plan_groupby, pg_plan_topn,
 * pg_plan_partition}; returns how many it would fill.  Bindings compare
 * against their own struct sizes at load time. */
int32_t pg_abi_struct_sizes(int32_t* out, int32_t n);

pg_status pg_page_serialize2(const pg_page* page, int32_t compress,
                             void* out, int64_t cap, int64_t* out_len);
/* parses and verifies; fills *out with malloc-backed host columns
 * (free with pg_page_free). F64 consumers reinterpret LONG_ARRAY bits. */
pg_status pg_page_deserialize(const void* buf, int64_t len, pg_page* out);
pg_status pg_page_free(pg_page* page);

#ifdef __cplusplus
}
#endif
#endif
