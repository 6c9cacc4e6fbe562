/*
 * oracle.h — CPU restatement of the reference hot path (TEST INFRASTRUCTURE).
 *
 * This library is the parity ORACLE for the GPU executor: a plain-C
 * restatement of Apache Cloudberry's segment-local scan→hash-join→hash-agg
 * pipeline semantics and its supporting formats/hashes, following the
 * reference sources cited per function in oracle.c.
 *
 * ONLY tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
 * call this library, and only as the checker / reported CPU baseline.  The
 * product path (cloudberry_amd + libgpuexec.so) never links or loads it and
 * fails loudly if the HIP extension is missing.
 */
#ifndef ORACLE_H
#define ORACLE_H
#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- bit-exact reference hashes (pinned against oracle/_ref) ---- */
uint32_t orc_hash_bytes_uint32(uint32_t k);
uint32_t orc_hashint8(int64_t v);
uint32_t orc_cdbhash_i64(int64_t v);                 /* 1-key cdbhash chain */
int32_t  orc_jump_consistent_hash(uint64_t key, int32_t nsegs);
int32_t  orc_route_i64(int64_t key, int32_t nsegs);  /* Motion routing */
void     orc_route_i64_batch(const int64_t *keys, int64_t n, int32_t nsegs,
                             int32_t *out);
uint32_t orc_hashint4(int32_t v);                    /* hashfunc.c:73-77 */
/* N-attribute cdbhash chain (cdbhash.c:171-247); types[k]: 0=int8, 1=int4;
 * NULL attributes contribute the rotation only */
uint32_t orc_cdbhash_multi(const int64_t *vals, const uint8_t *isnull,
                           const int32_t *types, int32_t nkeys);
void     orc_route_multi_batch(const int64_t *vals, const uint8_t *isnull,
                               const int32_t *types, int32_t nkeys, int64_t n,
                               int32_t nsegs, int32_t *out);
uint32_t orc_crc32c(uint32_t crc, const void *buf, size_t len); /* pg COMP_CRC32C state (no final xor) */

/* ---- date helpers (DateADT = int32 days since 2000-01-01) ---- */
int32_t orc_date_adt(int year, int month, int day);

/* ---- deterministic synthetic data (contract shared with HIP kernels) ---- */
int orc_set_threads(int n);   /* OpenMP thread count; returns effective */
uint64_t orc_splitmix64(uint64_t x);
uint64_t orc_mix(uint64_t seed, uint64_t stream, uint64_t idx);

typedef struct {
    int64_t *c_custkey;
    uint8_t *c_mktsegment;       /* 0..4; 0 = BUILDING */
    int64_t  n;
} orc_customer;

typedef struct {
    int64_t *o_orderkey;
    int64_t *o_custkey;
    int32_t *o_orderdate;
    int32_t *o_shippriority;
    int64_t  n;
} orc_orders;

typedef struct {
    int64_t *l_orderkey;
    double  *l_extendedprice;
    double  *l_discount;
    int32_t *l_shipdate;
    int64_t  n;
} orc_lineitem;

/* seg/nsegs shard by the reference's own distribution keys;
 * seg=0,nsegs=1 = whole table. Arrays are malloc'd; free with orc_free_*. */
int orc_gen_customer(double sf, uint64_t seed, int seg, int nsegs, orc_customer *out);
int orc_gen_orders  (double sf, uint64_t seed, int seg, int nsegs, orc_orders *out);
int orc_gen_lineitem(double sf, uint64_t seed, int seg, int nsegs, orc_lineitem *out);
void orc_free_customer(orc_customer *c);
void orc_free_orders(orc_orders *o);
void orc_free_lineitem(orc_lineitem *l);
/* table cardinalities for a scale factor (global, before sharding) */
int64_t orc_ncustomer(double sf);
int64_t orc_norders(double sf);

/* ---- AOCS column-store codec (byte-exact vs reference writer) ---- */
/* returns bytes written, or -1 on error.  width in {4,8}. */
int64_t orc_aocs_encode(const void *vals, int width, int64_t nrows,
                        int64_t first_rownum, int32_t blocksize,
                        uint8_t *out, int64_t outcap);
int64_t orc_aocs_encoded_size(int width, int64_t nrows, int32_t blocksize);
/* returns rows decoded, or -1 (bad header) / -2 (checksum mismatch). */
int64_t orc_aocs_decode(const uint8_t *stream, int64_t nbytes, int width,
                        void *out_vals, int64_t cap, int verify_checksums);
/* rows per full block for a fixed-width NOT NULL column */
int32_t orc_aocs_rows_per_block(int width, int32_t blocksize);

/* RLE_TYPE (Dense_Enhanced): exact restatement of the reference writer's
 * state machine, byte-exact with the compiled writer itself;
 * orc_aocs_decode handles Orig AND Dense(±RLE±DELTA) blocks transparently. */
int64_t orc_aocs_encode_rle(const void *vals, int width, int64_t nrows,
                            int64_t first_rownum, int32_t blocksize,
                            uint8_t *out, int64_t outcap);
/* RLE + DELTA_RANGE (full Dense_Enhanced feature set for NOT NULL int
 * columns): values within ±0x1FFFFFFF of the previous item are stored as
 * sign+magnitude varint deltas (datumstreamblock.c:2986, Reserved3 codec
 * datumstreamblock.h:790-930); repeats still RLE-compress. */
int64_t orc_aocs_encode_rle_delta(const void *vals, int width, int64_t nrows,
                                  int64_t first_rownum, int32_t blocksize,
                                  uint8_t *out, int64_t outcap);
/* zlib bulk compression (compresstype=zlib): per-block deflate of the Orig
 * content via compress2 (pg_compression.c:272-298); blocks that don't
 * shrink are stored uncompressed with compressedLength=0 (the reference's
 * own fallback).  orc_aocs_decode inflates transparently. */
int64_t orc_aocs_encode_zlib(const void *vals, int width, int64_t nrows,
                             int64_t first_rownum, int32_t blocksize,
                             int level, uint8_t *out, int64_t outcap);
/* zstd bulk compression (compresstype=zstd, gpcontrib/zstd — single-shot
 * ZSTD frames via ZSTD_compressCCtx/decompressDCtx).  The image ships
 * libzstd.so.1 without headers; prototypes declared locally. */
int64_t orc_aocs_encode_zstd(const void *vals, int width, int64_t nrows,
                             int64_t first_rownum, int32_t blocksize,
                             int level, uint8_t *out, int64_t outcap);
/* codec for bulk-compressed blocks: 1 = zlib (default), 2 = zstd */
int64_t orc_aocs_decode_c(const uint8_t *stream, int64_t nbytes, int width,
                          void *out_vals, int64_t cap, int verify_checksums,
                          int codec);
/* NULL-bearing columns: encoders take a per-row null flag array (NULL =
 * no nulls; both byte-exact vs the reference writer), the decoder fills
 * one validity byte per row (1 = non-null; null datums decode as zero).
 * orc_aocs_decode/_c refuse null-bearing blocks (-1). */
int64_t orc_aocs_encode_orig_nulls(const void *vals, const uint8_t *nulls,
                                   int width, int64_t nrows,
                                   int64_t first_rownum, int32_t blocksize,
                                   uint8_t *out, int64_t outcap);
int64_t orc_aocs_encode_rle_delta_nulls(const void *vals, const uint8_t *nulls,
                                        int width, int64_t nrows,
                                        int64_t first_rownum, int32_t blocksize,
                                        int delta,
                                        uint8_t *out, int64_t outcap);
int64_t orc_aocs_decode_nullable(const uint8_t *stream, int64_t nbytes,
                                 int width, void *out_vals,
                                 uint8_t *out_validity, int64_t cap,
                                 int verify_checksums, int codec);
/* varlena (text-like) Orig columns: payload = concatenated value bytes,
 * offsets[nrows+1] exclusive; short-form conversion and alignment follow
 * the reference writer exactly (byte-exact, see tests). */
int64_t orc_aocs_encode_varlena(const uint8_t *payload, const int64_t *offsets,
                                const uint8_t *nulls, int64_t nrows,
                                int64_t first_rownum, int32_t blocksize,
                                uint8_t *out, int64_t outcap);
int64_t orc_aocs_decode_varlena(const uint8_t *stream, int64_t nbytes,
                                int64_t nrows,
                                uint8_t *out_payload, int64_t payload_cap,
                                int64_t *out_offsets, uint8_t *out_validity,
                                int verify_checksums);
/* Dense_Enhanced rle_type varlena (RLE on repeated payloads) */
int64_t orc_aocs_encode_varlena_rle(const uint8_t *payload,
                                    const int64_t *offsets,
                                    const uint8_t *nulls, int64_t nrows,
                                    int64_t first_rownum, int32_t blocksize,
                                    uint8_t *out, int64_t outcap);

/* ---- Q3 pipeline (reference executor semantics) ---- */
typedef struct {
    int64_t l_orderkey;
    int32_t o_orderdate;
    int32_t o_shippriority;
    double  revenue;
    int64_t nitems;              /* lineitems folded into the group */
} orc_q3_group;

/* Runs scan→join→agg on already-materialized columns; returns group count
 * (sorted by l_orderkey asc) via *out (malloc'd; orc_free()).  cutoff =
 * DateADT of 1995-03-15 unless overridden. */
int64_t orc_q3(const orc_customer *c, const orc_orders *o,
               const orc_lineitem *l, int32_t cutoff, orc_q3_group **out);

/* numeric(15,2) mode: measures mapped to scaled int64 —
 * price in cents, discount in hundredths; revenue = Σ price_c·(100−disc_c),
 * an EXACT integer with implied scale 1e-4 (matches PG numeric for these
 * ranges).  Returns per-group numerators, sorted by l_orderkey. */
typedef struct {
    int64_t l_orderkey;
    int32_t o_orderdate;
    int32_t o_shippriority;
    int64_t revenue_num;         /* scale 1e-4 */
    int64_t nitems;
} orc_q3n_group;
int64_t orc_q3_numeric(const orc_customer *c, const orc_orders *o,
                       const orc_lineitem *l, int32_t cutoff,
                       orc_q3n_group **out);
void orc_free(void *p);

/* TPC-H Q1 core (BASELINE config 4): 6 fixed groups */
typedef struct {
    int8_t  returnflag;          /* 0..2 */
    int8_t  linestatus;          /* 0..1 */
    int64_t count;
    double  sum_price;
    double  sum_revenue;
} orc_q1_group;
int orc_q1(double sf, uint64_t seed, int32_t cutoff, orc_q1_group *out6);

#ifdef __cplusplus
}
#endif
#endif
