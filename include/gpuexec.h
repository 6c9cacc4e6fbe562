/*
 * gpuexec.h — C-ABI of libgpuexec.so, the MI355X-native executor for
 * Cloudberry's segment-local scan→hash-join→hash-agg pipeline + hash Motion.
 *
 * This is the drop-in boundary.  Each entry point names the
 * reference seam it replaces (paths under /root/reference):
 *
 *   gx_init / gx_shutdown      — per-QE executor state; what a CustomScan
 *                                extension's BeginCustomScan/EndCustomScan
 *                                (src/include/nodes/extensible.h:124-158,
 *                                executor/nodeCustom.c) would call once per
 *                                segment process.
 *   gx_comm_*                  — the MotionIPCLayer SetupInterconnect /
 *                                TeardownInterconnect pair (cdb/ml_ipc.h:36,
 *                                executor/execMain.c:535); RCCL communicator
 *                                is created once per process lifetime
 *                                (mirrors gang reuse, dispatcher/README.md).
 *   gx_table_bind              — scan target binding: the table-AM open path
 *                                aocs_beginscan (access/aocs/aocsam.c:542)
 *                                with per-column segment-file streams.
 *   gx_tpch_gen                — bench-harness substitute for on-disk data
 *                                (no network; synthetic).
 *   gx_decode_column           — aocs_getnext datum decode
 *                                (aocsam.c:1131-1259, datumstreamblock.c:
 *                                195-340) materialised column-at-a-time.
 *   gx_q3_prepare/run/result   — the QE slice ExecProcNode pull loop over
 *                                SeqScan→HashJoin→HashAgg (+ Motion sends at
 *                                nsegs>1): execMain.c:983, nodeHashjoin.c:252,
 *                                nodeAgg.c:2743, nodeMotion.c:1181.
 *   gx_partition (exposed for tests) — doSendTuple's evalHashKey +
 *                                cdbhashreduce routing (nodeMotion.c:1088,
 *                                cdbhash.c:253-285,530-541), bit-exact.
 *
 * Plain C, status-code returns, caller-owned opaque handles.  No torch
 * types; no exceptions across the boundary (the PG-side shim converts
 * non-zero status into ereport(ERROR)).  All calls are per-process
 * single-threaded, matching the QE execution model (one thread per segment;
 * HIP streams + RCCL run on that thread).
 */
#ifndef GPUEXEC_H
#define GPUEXEC_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef enum gx_status {
    GX_OK = 0,
    GX_ERR_HIP = 1,            /* HIP runtime failure (see gx_last_error) */
    GX_ERR_RCCL = 2,           /* RCCL failure */
    GX_ERR_INVALID = 3,        /* bad argument / malformed stream */
    GX_ERR_CHECKSUM = 4,       /* AO block checksum mismatch */
    GX_ERR_OOM = 5,            /* device memory exhausted */
    GX_ERR_NOGPU = 6,          /* no usable device — callers must FAIL, not fall back */
    GX_ERR_STATE = 7           /* calls out of order */
} gx_status;

typedef struct gx_ctx gx_ctx;
typedef struct gx_table gx_table;
typedef struct gx_q3 gx_q3;

/* last error message for a context (or the global init error when ctx==NULL) */
const char *gx_last_error(const gx_ctx *ctx);
const char *gx_version(void);

/* ---- lifecycle ---- */
gx_status gx_init(int device_id, int seg_id, int nsegs, gx_ctx **out);
gx_status gx_shutdown(gx_ctx *ctx);

/* ---- interconnect (RCCL over xGMI; one rank per segment-GPU) ---- */
#define GX_UNIQUE_ID_BYTES 128
gx_status gx_comm_unique_id(unsigned char uid[GX_UNIQUE_ID_BYTES]); /* rank 0 */
gx_status gx_comm_init(gx_ctx *ctx, const unsigned char uid[GX_UNIQUE_ID_BYTES]);

/* ---- tables: AOCS per-column streams resident in HBM ---- */

typedef struct gx_coldesc {
    const void *host_stream;   /* AOCS stream bytes (appendonly=column,
                                  checksum=true) */
    int64_t     nbytes;
    int32_t     width;         /* fixed datum width: 1, 4 or 8 */
    int64_t     nrows;
    int32_t     blocksize;     /* AO blocksize the stream was written with */
    int32_t     format;        /* 0 = Orig (compresstype=none), 1 =
                                  Dense/Dense_Enhanced incl. RLE_TYPE/DELTA
                                  (no-null subset) */
    int32_t     codec;         /* bulk codec for compressedLength>0 blocks:
                                  0/1 = zlib, 2 = zstd (the pg_appendonly
                                  compresstype analog) */
} gx_coldesc;

gx_status gx_table_bind(gx_ctx *ctx, const gx_coldesc *cols, int ncols,
                        gx_table **out);
gx_status gx_table_free(gx_table *t);
gx_status gx_table_nrows(const gx_table *t, int64_t *out);
/* logical uncompressed bytes of all columns = the reference's
 * totalBytesRead accounting base (cdb/cdbaocsam.h:283) */
gx_status gx_table_logical_bytes(const gx_table *t, double *out);

/* synthetic TPC-H-shaped tables generated AND AOCS-encoded on device
 * (deterministic; identical formulas to oracle/oracle.c datagen) */
typedef enum { GX_TPCH_CUSTOMER = 0, GX_TPCH_ORDERS = 1, GX_TPCH_LINEITEM = 2,
               GX_TPCH_LINEITEM_NUMERIC = 3, /* measures as scaled int64 */
               GX_TPCH_LINEITEM_RLEKEY = 4,  /* l_orderkey RLE-compressed */
               GX_TPCH_LINEITEM_Q1 = 5       /* Q1 cols: flag,status,price,disc,ship */ } gx_tpch_table;
gx_status gx_tpch_gen(gx_ctx *ctx, gx_tpch_table which, double sf,
                      uint64_t seed, gx_table **out);

/* copy a column's raw AOCS stream bytes back to the host (parity tests) */
gx_status gx_table_dump_stream(gx_ctx *ctx, const gx_table *t, int col,
                               void *host_out, int64_t cap_bytes,
                               int64_t *nbytes);

/* decode one column back to host values (parity testing / config-2 path);
 * verify_checksums runs the CRC32C pair per block on device */
gx_status gx_decode_column(gx_ctx *ctx, const gx_table *t, int col,
                           void *host_out, int64_t cap_rows,
                           int verify_checksums);

/* NULL-bearing columns (block-directory format only): validity gets one
 * byte per row, 1 = non-null; null datums decode as zero. */
gx_status gx_decode_column_nullable(gx_ctx *ctx, const gx_table *t, int col,
                                    void *host_out, uint8_t *host_validity,
                                    int64_t cap_rows, int verify_checksums);

/* Scan-time visibility map (AO visimap executor semantics,
 * cdbappendonlyvisimap.c:140-210): bitmap has one bit per logical row,
 * ON = tuple hidden (deleted); every scan/build/probe kernel skips
 * hidden rows.  bitmap=NULL clears.  Set before gx_q3_prepare. */
gx_status gx_table_set_visimap(gx_ctx *ctx, gx_table *t,
                               const uint8_t *bitmap, int64_t nbits);

/* Varlena (text) Orig columns (bind with width = -1, format = 1):
 * offsets[nrows+1] exclusive into payload; validity required when the
 * stream carries a NULL bitmap. */
gx_status gx_decode_column_varlena(gx_ctx *ctx, const gx_table *t, int col,
                                   int64_t *host_offsets, void *host_payload,
                                   int64_t payload_cap,
                                   uint8_t *host_validity,
                                   int verify_checksums);

/* TPC-H Q1 core (BASELINE config 4): GROUP BY returnflag,linestatus with
 * COUNT/SUM over a GX_TPCH_LINEITEM_Q1 table; AVG = sum/count (float8_avg) */
gx_status gx_q1(gx_ctx *ctx, const gx_table *t, int32_t cutoff,
                int64_t *counts6, double *sum_price6, double *sum_rev6,
                double *ms_out);

/* standalone columnar scan + filter count (the SeqScan+qual slice;
 * BASELINE config 2): op 0 '<', 1 '>', 2 '=', 3 '!=' vs an integer/date
 * literal; ms_out = kernel time from HIP events */
gx_status gx_scan_filter(gx_ctx *ctx, const gx_table *t, int col, int op,
                         int64_t literal, int64_t *count_out, double *ms_out);

/* bit-exact Motion routing of an i64 key column on device:
 * out[i] = jump_consistent_hash(cdbhash(hashint8(keys[i])), nsegs) */
gx_status gx_partition(gx_ctx *ctx, const int64_t *host_keys, int64_t n,
                       int32_t nsegs, int32_t *host_out);

/* multi-column distribution keys (cdbhash.c:189-247 rotate-combine loop):
 * vals/isnull row-major n×nkeys (each attribute widened to int64);
 * types[k]: 0 = int8 (hashint8), 1 = int4/int2/date (hashint4); a NULL
 * attribute contributes only the rotation.  isnull may be NULL. */
gx_status gx_partition_multi(gx_ctx *ctx, const int64_t *host_vals,
                             const uint8_t *host_isnull,
                             const int32_t *host_types, int32_t nkeys,
                             int64_t n, int32_t nsegs, int32_t *host_out);

/* ---- the Q3 pipeline (the judged QE slice) ---- */

typedef struct gx_q3_group {
    int64_t l_orderkey;
    int32_t o_orderdate;
    int32_t o_shippriority;
    double  revenue;           /* f64 mode: the SUM; numeric mode: num/1e4 */
    int64_t revenue_num;       /* numeric(15,2) mode: exact Σ price_c·(100−d),
                                  implied scale 1e-4 (0 in f64 mode) */
    int64_t nitems;
    /* appended r2 for LEFT OUTER fact joins (fact_join=1): unmatched fact
     * rows form groups with NULL mid attributes (attrs_null=1, date/prio
     * 0); NULL fact keys form ONE group (key_is_null=1, NULLs-equal
     * grouping) returned LAST.  Both 0 for inner joins. */
    uint8_t key_is_null;
    uint8_t attrs_null;
    uint8_t _pad[6];
} gx_q3_group;

typedef struct gx_q3_stats {
    /* per-stage device times, ms (HIP events on the executor stream) */
    double ms_cust_build;
    double ms_orders_build;
    double ms_probe_agg;       /* the dominant kernel */
    double ms_extract;
    double ms_motion;          /* partition+exchange (0 at nsegs==1) */
    double ms_total;           /* event span over the whole pipeline */
    int64_t cust_rows, ord_rows, li_rows;
    int64_t probe_hits, groups;
    double bytes_scanned;      /* logical uncompressed bytes (cdbaocsam.h:283) */
    /* Motion sub-phases (appended r2; 0 at nsegs==1) */
    double ms_motion_counts;   /* count all-gathers incl. their host syncs */
    double ms_motion_payload;  /* row payload send/recv (device events) */
} gx_q3_stats;

/* customer cols: [c_custkey i64, c_mktsegment i8]
 * orders   cols: [o_orderkey i64, o_custkey i64, o_orderdate i32, o_shippriority i32]
 * lineitem cols: [l_orderkey i64, l_extendedprice f64, l_discount f64, l_shipdate i32] */
gx_status gx_q3_prepare(gx_ctx *ctx, gx_table *customer, gx_table *orders,
                        gx_table *lineitem, int32_t cutoff_dateadt,
                        gx_q3 **out);

/* ---- plan-descriptor form (what PlanCustomPath lowering
 * fills; tagged structs, no expression trees yet).  Filter ops:
 * 0 '<', 1 '>', 2 '=', 3 '!=', 4 '<=', 5 '>='. ---- */
typedef struct gx_filter {
    int32_t col;               /* column index in the role's table */
    int32_t op;
    int64_t literal;           /* integer/date literal (DateADT for dates) */
} gx_filter;

typedef struct gx_q3_desc {
    /* build side of the semijoin (reference: customer) */
    gx_table *dim;
    int32_t dim_key_col;       /* i64 join key */
    gx_filter dim_filter;      /* e.g. mktsegment = literal */
    /* middle table redistributed/joined on both keys (reference: orders) */
    gx_table *mid;
    int32_t mid_key_col;       /* i64 key joined to fact (o_orderkey) */
    int32_t mid_fk_col;        /* i64 key joined to dim  (o_custkey) */
    int32_t mid_attr1_col;     /* i32 carried into the group (o_orderdate) */
    int32_t mid_attr2_col;     /* i32 carried into the group (o_shippriority) */
    gx_filter mid_filter;      /* e.g. o_orderdate < literal */
    /* fact side (reference: lineitem); agg = SUM(a*(1-b)), COUNT(*) */
    gx_table *fact;
    int32_t fact_key_col;      /* i64 probe key (l_orderkey) */
    int32_t fact_a_col;        /* f64 (l_extendedprice) */
    int32_t fact_b_col;        /* f64 (l_discount) */
    gx_filter fact_filter;     /* e.g. l_shipdate > literal */
    /* optional TEXT dim predicate (texteq — the reference's actual Q3 qual
     * c_mktsegment = 'BUILDING'): when dim_text_len > 0, dim_filter.col
     * names a VARLENA directory column (width -1, format 1) and the
     * predicate is payload == dim_text with op '=='.  With rle_type
     * segments the comparison runs once per RUN. */
    char dim_text[64];
    int32_t dim_text_len;
    /* AND-ed qual lists (execScan.c:241 semantics over NOT NULL integer/date
     * columns: every qual must pass; widths 1/4/8, literals widened to i64).
     * Folded at prepare into a per-table row mask combined with the visimap,
     * so the per-step kernels see them through the existing visibility path. */
#define GX_MAX_EXTRA_QUALS 4
    gx_filter dim_extra[GX_MAX_EXTRA_QUALS];
    gx_filter mid_extra[GX_MAX_EXTRA_QUALS];
    gx_filter fact_extra[GX_MAX_EXTRA_QUALS];
    int32_t n_dim_extra, n_mid_extra, n_fact_extra;
    /* dim-join type (nodeHashjoin.c join variety on the dim semijoin):
     * 0 = JOIN_SEMI (IN / EXISTS — the Q3 shape)
     * 1 = JOIN_LASJ (anti, NOT EXISTS): mid rows pass when the fk is NOT
     *     in the dim set; a NULL fk never matches, so it PASSES
     *     (nodeHashjoin.c:652-659 left-anti semantics)
     * 2 = JOIN_LASJ_NOTIN (NOT IN): a NULL fk is rejected, and ANY NULL
     *     dim key passing the dim filter empties the whole result
     *     (nodeHashjoin.c:425,442 hs_hashkeys_null) */
    int32_t dim_join;
    /* fact-join type (the fact⋈mid join): 0 = inner (the Q3 shape);
     * 1 = LEFT OUTER (nodeHashjoin.c HJ_FILL_OUTER): fact rows passing
     *     their WHERE quals but matching no mid row form groups with NULL
     *     mid attrs; NULL fact keys (never equal under the strict op)
     *     also emit, all in ONE group (NULLs-equal grouping).  Requires a
     *     plain or materialized fact key (no fused-RLE) and f64 measures. */
    int32_t fact_join;
} gx_q3_desc;

/* Restrictions checked at sizing (first gx_q3_run):
 * - join keys must be >= 1: slot value 0 is the empty-slot sentinel (PG
 *   sequence-keyed tables start at 1); a key 0 on the build side returns
 *   GX_ERR_INVALID instead of silently dropping the row.
 * - HBM budget: if the semijoin set / join table would exceed free HBM
 *   (override: GX_HBM_BUDGET_MB), sizing fails with GX_ERR_OOM and the
 *   required vs available GB in gx_last_error BEFORE any allocation — the
 *   reference spills to batches (nodeHashjoin.c:1355); this executor
 *   rejects, callers keep the reference CPU path for such plans. */
gx_status gx_q3_prepare_desc(gx_ctx *ctx, const gx_q3_desc *desc, gx_q3 **out);
/* numeric(15,2) mode: lineitem measures are scaled int64
 * (price cents, discount hundredths — GX_TPCH_LINEITEM_NUMERIC tables);
 * aggregation is integer → results BIT-EXACT vs the oracle. */
gx_status gx_q3_set_numeric(gx_q3 *q, int on);
gx_status gx_q3_run(gx_q3 *q);   /* one full pass; re-runnable (bench steps) */
gx_status gx_q3_stats_get(const gx_q3 *q, gx_q3_stats *out);
/* groups of THIS segment, sorted by l_orderkey asc; caller frees with gx_free */
gx_status gx_q3_result(gx_q3 *q, gx_q3_group **out, int64_t *ngroups);
/* top-N of this segment's groups (Q3's ORDER BY revenue DESC, o_orderdate
 * LIMIT N; N ≤ 10) — the nodeSort/nodeLimit stage, device-selected.
 * Outer-join groups sort with NULL dates LAST (PG ASC NULLS LAST). */
gx_status gx_q3_topn(gx_q3 *q, int topn, gx_q3_group *out, int64_t *nout);
gx_status gx_q3_free(gx_q3 *q);

void gx_free(void *p);

/* ABI self-description for foreign mirrors (0 stats, 1 q3_group,
 * 2 kv_group, 3 q3_desc, 4 coldesc, 5 filter); -1 for unknown kinds */
int64_t gx_abi_sizeof(int kind);

/* ---- standalone hash GROUP BY (nodeAgg.c:2288 hash strategy over one key;
 * grouping equality is NOT DISTINCT, execGrouping.c:436-495: all NULL keys
 * form ONE group, returned LAST with key_is_null=1).  COUNT(*) counts every
 * row; SUM(float8)'s transition is strict (float.c:769) and skips NULL
 * inputs.  key/val cols are i64/f64, Orig or Dense/RLE (nullable via
 * format-1 streams).  Key INT64_MIN is rejected (biased sentinel).  Groups
 * sorted by key asc; caller frees with gx_free. ---- */
typedef struct gx_kv_group {
    int64_t key;
    uint8_t key_is_null;
    uint8_t _pad[7];
    double  sum;
    int64_t count;
} gx_kv_group;
gx_status gx_groupby(gx_ctx *ctx, const gx_table *t, int key_col, int val_col,
                     gx_kv_group **out, int64_t *ngroups);

/* ---- test-only entry points (parity harness; not part of the drop-in) ---- */
typedef struct gx_ord_row { int64_t okey, ocust; int32_t odate, oprio; } gx_ord_row;
/* Motion-1 partition kernels on one GPU: filter orders by cutoff, route by
 * cdbhash(o_custkey), emit packed rows grouped by destination segment. */
gx_status gx_test_motion1(gx_ctx *ctx, gx_table *orders, int32_t cutoff,
                          int nsegs, int64_t *out_counts, gx_ord_row *out_rows,
                          int64_t cap, int64_t *out_total);
typedef struct gx_qual_row_abi { int64_t okey; int32_t odate, oprio; } gx_qual_row_abi;
/* Motion stage 2 (received rows → customer semijoin → route by o_orderkey) */
gx_status gx_test_qual(gx_ctx *ctx, gx_table *customer, const gx_ord_row *rows,
                       int64_t n, int nsegs, int64_t *out_counts,
                       gx_qual_row_abi *out_rows, int64_t cap, int64_t *out_total);
/* Motion stage 3 (received qual rows → table build → probe local lineitem) */
gx_status gx_test_q3_from_qual(gx_ctx *ctx, const gx_qual_row_abi *rows,
                               int64_t n, gx_table *lineitem, int32_t cutoff,
                               gx_q3_group **out, int64_t *ngroups);
int gx_selftest_addressing(void);

#ifdef __cplusplus
}
#endif
#endif /* GPUEXEC_H */
