/*
 * oracle.c — CPU restatement of the Cloudberry hot path (TEST INFRASTRUCTURE).
 *
 * Each function cites the reference file:line it follows (paths relative to
 * /root/reference).  See oracle.h for the usage contract: this library is
 * the parity checker and reported CPU baseline only — never the product path.
 */
#include "oracle.h"
#include <stdlib.h>
#include <string.h>
#include <math.h>
#include <zlib.h>

/* libzstd.so.1 is present without zstd.h — declare the single-shot API */
extern size_t ZSTD_compress(void *dst, size_t dstCap, const void *src,
                            size_t srcSize, int level);
extern size_t ZSTD_decompress(void *dst, size_t dstCap, const void *src,
                              size_t srcSize);
extern unsigned ZSTD_isError(size_t code);
extern size_t ZSTD_compressBound(size_t srcSize);
#ifdef _OPENMP
#include <omp.h>
#endif

int orc_set_threads(int n)
{
#ifdef _OPENMP
    if (n > 0) omp_set_num_threads(n);
    return omp_get_max_threads();
#else
    (void) n;
    return 1;
#endif
}

/* ======================================================================
 * Hashing — bit-exact restatements
 * ====================================================================== */

static inline uint32_t rot32(uint32_t x, int k) { return (x << k) | (x >> (32 - k)); }

/*
 * Jenkins final() — src/common/hashfn.c:133-142.
 */
#define ORC_FINAL(a,b,c) \
{ \
  c ^= b; c -= rot32(b,14); \
  a ^= c; a -= rot32(c,11); \
  b ^= a; b -= rot32(a,25); \
  c ^= b; c -= rot32(b,16); \
  a ^= c; a -= rot32(c, 4); \
  b ^= a; b -= rot32(a,14); \
  c ^= b; c -= rot32(b,24); \
}

/* hash_bytes_uint32 — src/common/hashfn.c:620-637 */
uint32_t orc_hash_bytes_uint32(uint32_t k)
{
    uint32_t a, b, c;
    a = b = c = 0x9e3779b9 + (uint32_t) sizeof(uint32_t) + 3923095;
    a += k;
    ORC_FINAL(a, b, c);
    return c;
}

/* hashint8 — src/backend/access/hash/hashfunc.c:85-101 */
uint32_t orc_hashint8(int64_t val)
{
    uint32_t lohalf = (uint32_t) val;
    uint32_t hihalf = (uint32_t) ((uint64_t) val >> 32);
    lohalf ^= (val >= 0) ? hihalf : ~hihalf;
    return orc_hash_bytes_uint32(lohalf);
}

/*
 * cdbhash for a single not-null int8 key —
 * src/backend/cdb/cdbhash.c:171-247: cdbhashinit (hash=0, non-legacy) then
 * per attribute: rotate-left-1 then XOR the type hash (hashint8 here).
 */
uint32_t orc_cdbhash_i64(int64_t v)
{
    uint32_t hashkey = 0;                                  /* cdbhashinit */
    hashkey = (hashkey << 1) | ((hashkey & 0x80000000u) ? 1 : 0);
    hashkey ^= orc_hashint8(v);
    return hashkey;
}

/* jump_consistent_hash — src/backend/cdb/cdbhash.c:530-541 (Lamping-Veach) */
int32_t orc_jump_consistent_hash(uint64_t key, int32_t num_segments)
{
    int64_t b = -1;
    int64_t j = 0;
    while (j < num_segments)
    {
        b = j;
        key = key * 2862933555777941757ULL + 1;
        j = (int64_t) ((double) (b + 1) *
                       ((double) (1LL << 31) / (double) ((key >> 33) + 1)));
    }
    return (int32_t) b;
}

/*
 * Motion / DISTRIBUTED BY routing for an int8 key —
 * nodeMotion.c:1088 evalHashKey → cdbhashreduce (cdbhash.c:253-285),
 * REDUCE_JUMP_HASH branch (the non-legacy default).
 */
int32_t orc_route_i64(int64_t key, int32_t nsegs)
{
    return orc_jump_consistent_hash((uint64_t) orc_cdbhash_i64(key), nsegs);
}

void orc_route_i64_batch(const int64_t *keys, int64_t n, int32_t nsegs, int32_t *out)
{
    for (int64_t i = 0; i < n; i++)
        out[i] = orc_route_i64(keys[i], nsegs);
}

/* hashint4 — src/backend/access/hash/hashfunc.c:73-77 (hash_uint32 of the
 * int32 value; int2/int4/int8 produce compatible hashes for equal values) */
uint32_t orc_hashint4(int32_t val)
{
    return orc_hash_bytes_uint32((uint32_t) val);
}

/*
 * cdbhash over N attributes in declared order —
 * src/backend/cdb/cdbhash.c:171-247: cdbhashinit (hash = 0, non-legacy),
 * then per attribute
 *     hashkey = rotate_left_1(hashkey);
 *     if (!isnull) hashkey ^= typehash(val);
 * A NULL attribute contributes ONLY the rotation (cdbhash.c:195-216).
 * types[k]: 0 = int8 (hashint8), 1 = int4/int2/date (hashint4).
 * vals carries each attribute widened to int64 (sign-preserving).
 */
uint32_t orc_cdbhash_multi(const int64_t *vals, const uint8_t *isnull,
                           const int32_t *types, int32_t nkeys)
{
    uint32_t hashkey = 0;                                  /* cdbhashinit */
    for (int32_t k = 0; k < nkeys; k++)
    {
        hashkey = (hashkey << 1) | ((hashkey & 0x80000000u) ? 1 : 0);
        if (isnull && isnull[k]) continue;
        hashkey ^= (types && types[k] == 1)
                       ? orc_hashint4((int32_t) vals[k])
                       : orc_hashint8(vals[k]);
    }
    return hashkey;
}

/* multi-key Motion routing (row-major vals/isnull, n rows × nkeys) */
void orc_route_multi_batch(const int64_t *vals, const uint8_t *isnull,
                           const int32_t *types, int32_t nkeys, int64_t n,
                           int32_t nsegs, int32_t *out)
{
    for (int64_t i = 0; i < n; i++)
        out[i] = orc_jump_consistent_hash(
            (uint64_t) orc_cdbhash_multi(vals + i * nkeys,
                                         isnull ? isnull + i * nkeys : 0,
                                         types, nkeys),
            nsegs);
}

/*
 * CRC-32C in PostgreSQL COMP_CRC32C semantics (reflected Castagnoli,
 * init 0xFFFFFFFF, NO final inversion — "by historical accident" the AO
 * block checksums are not inverted, cdbappendonlystorageformat.c:41-47).
 */
static uint32_t crc32c_table[256];
static int crc32c_ready = 0;
static void crc32c_init(void)
{
    for (uint32_t i = 0; i < 256; i++)
    {
        uint32_t c = i;
        for (int k = 0; k < 8; k++)
            c = (c & 1) ? (0x82F63B78u ^ (c >> 1)) : (c >> 1);
        crc32c_table[i] = c;
    }
    crc32c_ready = 1;
}
uint32_t orc_crc32c(uint32_t crc, const void *buf, size_t len)
{
    const uint8_t *p = (const uint8_t *) buf;
    if (!crc32c_ready) crc32c_init();
    while (len--)
        crc = crc32c_table[(crc ^ *p++) & 0xFF] ^ (crc >> 8);
    return crc;
}

/* ======================================================================
 * Dates — DateADT is int32 days since 2000-01-01 (src/include/utils/date.h:23,
 * POSTGRES_EPOCH_JDATE datatype/timestamp.h:209).
 * ====================================================================== */

/* days_from_civil (Howard Hinnant's algorithm), shifted to the 2000-01-01 epoch */
int32_t orc_date_adt(int y, int m, int d)
{
    int64_t yy = y - (m <= 2);
    int64_t era = (yy >= 0 ? yy : yy - 399) / 400;
    unsigned yoe = (unsigned) (yy - era * 400);
    unsigned doy = (153u * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1;
    unsigned doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
    int64_t days_unix = era * 146097 + (int64_t) doe - 719468;   /* days since 1970-01-01 */
    return (int32_t) (days_unix - 10957);                        /* 2000-01-01 is unix day 10957 */
}

/* ======================================================================
 * Deterministic synthetic TPC-H-shaped data (SURVEY §8d distributions).
 * THE CONTRACT: identical formulas in cloudberry_amd/csrc/gx_kernels.hip.
 * ====================================================================== */

uint64_t orc_splitmix64(uint64_t x)
{
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

uint64_t orc_mix(uint64_t seed, uint64_t stream, uint64_t idx)
{
    return orc_splitmix64(orc_splitmix64(seed ^ (stream * 0xA24BAED4963EE407ULL)) + idx);
}

/* stream ids (shared constants) */
enum {
    ST_CUST_SEG = 1,
    ST_ORD_CUST = 2,
    ST_ORD_DATE = 3,
    ST_ORD_PRIO = 4,
    ST_LI_COUNT = 5,
    ST_LI_SHIP  = 6,
    ST_LI_PRICE = 7,
    ST_LI_DISC  = 8,
    ST_LI_FLAG  = 9,
    ST_LI_STATUS = 10,
};

int64_t orc_ncustomer(double sf) { return (int64_t) (150000.0 * sf + 0.5); }
int64_t orc_norders(double sf)   { return (int64_t) (1500000.0 * sf + 0.5); }

/* date ranges (TPC-H 4.3.2/4.3.3): orders 1992-01-01..1998-08-02,
 * lineitem shipdate 1992-01-02..1998-12-01 */
#define ORD_DATE_LO() orc_date_adt(1992, 1, 1)
#define ORD_DATE_SPAN 2405   /* inclusive span-1; lo + mix%(span+1) */
#define LI_DATE_LO()  orc_date_adt(1992, 1, 2)
#define LI_DATE_SPAN  2525

static inline uint8_t gen_mktsegment(uint64_t seed, int64_t i)
{ return (uint8_t) (orc_mix(seed, ST_CUST_SEG, (uint64_t) i) % 5); }

static inline int64_t gen_ocustkey(uint64_t seed, int64_t i, int64_t ncust)
{
    int64_t pool = (ncust * 2) / 3;            /* 1/3 of customers orderless */
    if (pool < 1) pool = 1;
    return 1 + (int64_t) (orc_mix(seed, ST_ORD_CUST, (uint64_t) i) % (uint64_t) pool);
}
static inline int32_t gen_odate(uint64_t seed, int64_t i)
{ return ORD_DATE_LO() + (int32_t) (orc_mix(seed, ST_ORD_DATE, (uint64_t) i) % (ORD_DATE_SPAN + 1)); }
static inline int32_t gen_oprio(uint64_t seed, int64_t i)
{ return (int32_t) (orc_mix(seed, ST_ORD_PRIO, (uint64_t) i) % 5); }

static inline int32_t gen_nlines(uint64_t seed, int64_t okey)
{ return 1 + (int32_t) (orc_mix(seed, ST_LI_COUNT, (uint64_t) okey) % 7); }
static inline int32_t gen_shipdate(uint64_t seed, int64_t okey, int32_t line)
{ return LI_DATE_LO() + (int32_t) (orc_mix(seed, ST_LI_SHIP, (uint64_t) okey * 8 + line) % (LI_DATE_SPAN + 1)); }
static inline double gen_price(uint64_t seed, int64_t okey, int32_t line)
{ return (double) (90000 + orc_mix(seed, ST_LI_PRICE, (uint64_t) okey * 8 + line) % 10410001ULL) / 100.0; }
static inline uint8_t gen_returnflag(uint64_t seed, int64_t okey, int32_t line)
{ return (uint8_t) (orc_mix(seed, ST_LI_FLAG, (uint64_t) okey * 8 + line) % 3); }
static inline uint8_t gen_linestatus(uint64_t seed, int64_t okey, int32_t line)
{ return (uint8_t) (orc_mix(seed, ST_LI_STATUS, (uint64_t) okey * 8 + line) % 2); }
static inline double gen_discount(uint64_t seed, int64_t okey, int32_t line)
{ return (double) (orc_mix(seed, ST_LI_DISC, (uint64_t) okey * 8 + line) % 11) / 100.0; }

int orc_gen_customer(double sf, uint64_t seed, int seg, int nsegs, orc_customer *out)
{
    int64_t n = orc_ncustomer(sf), kept = 0;
    int64_t cap = (nsegs == 1) ? n : (n / nsegs + (int64_t) (4.0 * sqrt((double) n / nsegs)) + 64);
    out->c_custkey = malloc(sizeof(int64_t) * cap);
    out->c_mktsegment = malloc(cap);
    if (!out->c_custkey || !out->c_mktsegment) return -1;
    for (int64_t i = 0; i < n; i++)
    {
        int64_t key = i + 1;
        if (nsegs > 1 && orc_route_i64(key, nsegs) != seg) continue;
        if (kept == cap)
        {
            cap = cap * 2;
            out->c_custkey = realloc(out->c_custkey, sizeof(int64_t) * cap);
            out->c_mktsegment = realloc(out->c_mktsegment, cap);
        }
        out->c_custkey[kept] = key;
        out->c_mktsegment[kept] = gen_mktsegment(seed, i);
        kept++;
    }
    out->n = kept;
    return 0;
}

int orc_gen_orders(double sf, uint64_t seed, int seg, int nsegs, orc_orders *out)
{
    int64_t n = orc_norders(sf), ncust = orc_ncustomer(sf), kept = 0;
    int64_t cap = (nsegs == 1) ? n : (n / nsegs + (int64_t) (4.0 * sqrt((double) n / nsegs)) + 64);
    out->o_orderkey = malloc(sizeof(int64_t) * cap);
    out->o_custkey = malloc(sizeof(int64_t) * cap);
    out->o_orderdate = malloc(sizeof(int32_t) * cap);
    out->o_shippriority = malloc(sizeof(int32_t) * cap);
    for (int64_t i = 0; i < n; i++)
    {
        int64_t key = i + 1;
        if (nsegs > 1 && orc_route_i64(key, nsegs) != seg) continue;
        if (kept == cap)
        {
            cap *= 2;
            out->o_orderkey = realloc(out->o_orderkey, sizeof(int64_t) * cap);
            out->o_custkey = realloc(out->o_custkey, sizeof(int64_t) * cap);
            out->o_orderdate = realloc(out->o_orderdate, sizeof(int32_t) * cap);
            out->o_shippriority = realloc(out->o_shippriority, sizeof(int32_t) * cap);
        }
        out->o_orderkey[kept] = key;
        out->o_custkey[kept] = gen_ocustkey(seed, i, ncust);
        out->o_orderdate[kept] = gen_odate(seed, i);
        out->o_shippriority[kept] = gen_oprio(seed, i);
        kept++;
    }
    out->n = kept;
    return 0;
}

int orc_gen_lineitem(double sf, uint64_t seed, int seg, int nsegs, orc_lineitem *out)
{
    int64_t nord = orc_norders(sf), kept = 0;
    int64_t cap = (nsegs == 1) ? nord * 4 + 64 : (nord * 4 / nsegs + (int64_t) (8.0 * sqrt((double) nord * 4 / nsegs)) + 64);
    out->l_orderkey = malloc(sizeof(int64_t) * cap);
    out->l_extendedprice = malloc(sizeof(double) * cap);
    out->l_discount = malloc(sizeof(double) * cap);
    out->l_shipdate = malloc(sizeof(int32_t) * cap);
    for (int64_t o = 1; o <= nord; o++)
    {
        if (nsegs > 1 && orc_route_i64(o, nsegs) != seg) continue;
        int32_t nl = gen_nlines(seed, o);
        if (kept + nl > cap)
        {
            cap = cap * 2 + nl;
            out->l_orderkey = realloc(out->l_orderkey, sizeof(int64_t) * cap);
            out->l_extendedprice = realloc(out->l_extendedprice, sizeof(double) * cap);
            out->l_discount = realloc(out->l_discount, sizeof(double) * cap);
            out->l_shipdate = realloc(out->l_shipdate, sizeof(int32_t) * cap);
        }
        for (int32_t j = 0; j < nl; j++)
        {
            out->l_orderkey[kept] = o;
            out->l_extendedprice[kept] = gen_price(seed, o, j);
            out->l_discount[kept] = gen_discount(seed, o, j);
            out->l_shipdate[kept] = gen_shipdate(seed, o, j);
            kept++;
        }
    }
    out->n = kept;
    return 0;
}

void orc_free_customer(orc_customer *c)
{ free(c->c_custkey); free(c->c_mktsegment); c->c_custkey = NULL; c->c_mktsegment = NULL; }
void orc_free_orders(orc_orders *o)
{ free(o->o_orderkey); free(o->o_custkey); free(o->o_orderdate); free(o->o_shippriority);
  o->o_orderkey = o->o_custkey = NULL; o->o_orderdate = o->o_shippriority = NULL; }
void orc_free_lineitem(orc_lineitem *l)
{ free(l->l_orderkey); free(l->l_extendedprice); free(l->l_discount); free(l->l_shipdate);
  l->l_orderkey = NULL; l->l_extendedprice = l->l_discount = NULL; l->l_shipdate = NULL; }
void orc_free(void *p) { free(p); }

/* ======================================================================
 * AOCS column-store codec — byte-exact vs the reference writer for
 * appendonly=column, compresstype=none, checksum=true, NOT NULL fixed-width.
 *
 * Stream = AO SmallContent blocks (cdbappendonlystorage_int.h:64-147):
 *   [0..8)   header bitfields  (headerKind=1, hasFirstRowNum=1,
 *            executorBlockKind=1 = AOCSBK_BLOCK datumstream.c:39-44)
 *   [8..12)  block CRC32C over [16, blockLen)   (cdbappendonlystorageformat.c:50-78)
 *   [12..16) header CRC32C over [0,12)          (cdbappendonlystorageformat.c:25-48)
 *   [16..24) firstRowNum int64                  (cdbappendonlystorageformat.c:89-123)
 *   [24..)   content: DatumStreamBlock_Orig 16B (datumstreamblock.h:73-84)
 *            {version=0,flags=0,ndatum,encrypted=0,nullsz=0,sz} then datums
 *            at MAXALIGN(16)=16 (datumstreamblock.c:273-279)
 *   whole block rounded up to 8 B (AOStorage_RoundUp8, cdbappendonlystorage.h:39)
 *
 * Capacity per block (DatumStreamBlockWrite_OrigHasSpace,
 * datumstreamblock.c:1508-1560 with maxDataBlockSize = blocksize −
 * AoHeader_Size(false,true,true)=24, datumstream.c:588-631):
 *   accept while 16 + n*width + width < blocksize-24  and  n+1 < 16383.
 * ====================================================================== */

static int64_t decode_dense_content(const uint8_t *c, int width,
                                    uint8_t *dst, int64_t cap_rows);
static int64_t decode_dense_content_v(const uint8_t *c, int width,
                                      uint8_t *dst, uint8_t *validity,
                                      int64_t cap_rows);

int32_t orc_aocs_rows_per_block(int width, int32_t blocksize)
{
    int32_t maxdata = blocksize - 24;
    int32_t n = 0;
    while ((n + 1 < 16383) && (16 + n * width + width < maxdata))
        n++;
    return n;                    /* rows in every full block */
}

int64_t orc_aocs_encoded_size(int width, int64_t nrows, int32_t blocksize)
{
    int32_t rpb = orc_aocs_rows_per_block(width, blocksize);
    int64_t nblocks = (nrows + rpb - 1) / rpb;
    int64_t sz = 0;
    for (int64_t b = 0; b < nblocks; b++)
    {
        int64_t rows = (b == nblocks - 1) ? (nrows - b * (int64_t) rpb) : rpb;
        int64_t content = 16 + rows * width;
        sz += (24 + content + 7) & ~7LL;
    }
    return sz;
}

static void put_u32le(uint8_t *p, uint32_t v) { memcpy(p, &v, 4); }

int64_t orc_aocs_encode(const void *vals, int width, int64_t nrows,
                        int64_t first_rownum, int32_t blocksize,
                        uint8_t *out, int64_t outcap)
{
    int32_t rpb = orc_aocs_rows_per_block(width, blocksize);
    const uint8_t *src = (const uint8_t *) vals;
    int64_t off = 0, row = 0;
    while (row < nrows)
    {
        int32_t rows = (int32_t) ((nrows - row < rpb) ? (nrows - row) : rpb);
        int32_t sz = rows * width;
        int32_t content = 16 + sz;
        int64_t blocklen = (24 + content + 7) & ~7LL;
        if (off + blocklen > outcap) return -1;
        uint8_t *blk = out + off;
        memset(blk, 0, blocklen);

        /* AOSmallContentHeader Init macros, cdbappendonlystorage_int.h:150-170 */
        uint32_t b03 = 0, b47 = 0;
        b03 |= (1u << 28);                       /* headerKind = SmallContent */
        b03 |= (1u << 27);                       /* hasFirstRowNum */
        b03 |= (1u << 24);                       /* executorBlockKind = AOCSBK_BLOCK */
        b03 |= (0x00FFFC00u & ((uint32_t) rows << 10));      /* rowCount 14b */
        b03 |= (((uint32_t) content >> 11) & 0x3FFu);        /* dataLength hi 10b */
        b47 |= (((uint32_t) content & 0x7FFu) << 21);        /* dataLength lo 11b */
        /* compressedLength = 0 */
        put_u32le(blk, b03);
        put_u32le(blk + 4, b47);

        int64_t frn = first_rownum + row;
        memcpy(blk + 16, &frn, 8);

        /* DatumStreamBlock_Orig, datumstreamblock.h:73-84 */
        uint8_t *content_p = blk + 24;
        int16_t v16;
        v16 = 0;              memcpy(content_p + 0, &v16, 2);   /* version = Original */
        v16 = 0;              memcpy(content_p + 2, &v16, 2);   /* flags (no null bitmap) */
        v16 = (int16_t) rows; memcpy(content_p + 4, &v16, 2);   /* ndatum */
        v16 = 0;              memcpy(content_p + 6, &v16, 2);   /* encrypted */
        int32_t v32 = 0;      memcpy(content_p + 8, &v32, 4);   /* nullsz */
        v32 = sz;             memcpy(content_p + 12, &v32, 4);  /* sz */
        memcpy(content_p + 16, src + row * (int64_t) width, sz);

        /* checksums: block CRC over [16, blockLen) incl. trailing zero pad,
         * then header CRC over [0,12)  (cdbappendonlystorageformat.c:160-190) */
        put_u32le(blk + 8, orc_crc32c(0xFFFFFFFFu, blk + 16, blocklen - 16));
        put_u32le(blk + 12, orc_crc32c(0xFFFFFFFFu, blk, 12));

        off += blocklen;
        row += rows;
    }
    return off;
}

int64_t orc_aocs_decode(const uint8_t *stream, int64_t nbytes, int width,
                        void *out_vals, int64_t cap, int verify_checksums)
{
    return orc_aocs_decode_c(stream, nbytes, width, out_vals, cap,
                             verify_checksums, 1);
}

int64_t orc_aocs_decode_c(const uint8_t *stream, int64_t nbytes, int width,
                          void *out_vals, int64_t cap, int verify_checksums,
                          int codec)
{
    return orc_aocs_decode_nullable(stream, nbytes, width, out_vals, NULL,
                                    cap, verify_checksums, codec);
}

/* like orc_aocs_decode_c but handles NULL-bearing blocks: out_validity gets
 * one byte per row (1 = non-null), null datums decode as zero.  With
 * out_validity == NULL any null-bearing block fails loudly (-1). */
int64_t orc_aocs_decode_nullable(const uint8_t *stream, int64_t nbytes,
                                 int width, void *out_vals,
                                 uint8_t *out_validity, int64_t cap,
                                 int verify_checksums, int codec)
{
    int64_t off = 0, row = 0;
    uint8_t *dst = (uint8_t *) out_vals;
    while (off + 24 <= nbytes)
    {
        uint32_t b03, b47;
        memcpy(&b03, stream + off, 4);
        memcpy(&b47, stream + off + 4, 4);
        if (b03 == 0 && b47 == 0) break;         /* zero padding tail */
        uint32_t kind = (b03 >> 28) & 7;
        uint32_t hasfrn = (b03 >> 27) & 1;
        uint32_t rows, datalen, complen = 0;
        if (kind == 1)                          /* SmallContent */
        {
            rows = (b03 & 0x00FFFC00u) >> 10;
            datalen = ((b03 & 0x3FFu) << 11) | ((b47 & 0xFFE00000u) >> 21);
            complen = b47 & 0x1FFFFFu;          /* bulk-compressed if >0 */
        }
        else if (kind == 3)                     /* NonBulkDenseContent */
        {
            rows = b47 & 0x3FFFFFFFu;
            datalen = b03 & 0x1FFFFFu;
        }
        else
            return -1;
        if (!hasfrn) return -1;
        int64_t blocklen = (24 + (int64_t) (complen ? complen : datalen) + 7) & ~7LL;
        if (off + blocklen > nbytes) return -1;
        if (verify_checksums)
        {
            uint32_t bc, hc;
            memcpy(&bc, stream + off + 8, 4);
            memcpy(&hc, stream + off + 12, 4);
            if (hc != orc_crc32c(0xFFFFFFFFu, stream + off, 12)) return -2;
            if (bc != orc_crc32c(0xFFFFFFFFu, stream + off + 16, blocklen - 16)) return -2;
        }
        const uint8_t *content = stream + off + 24;
        uint8_t *inflated = NULL;
        if (complen > 0)
        {
            inflated = malloc(datalen);
            if (codec == 2)
            {
                size_t r = ZSTD_decompress(inflated, datalen, content, complen);
                if (ZSTD_isError(r) || r != datalen)
                { free(inflated); return -1; }
            }
            else
            {
                unsigned long dl = datalen;
                if (uncompress(inflated, &dl, content, complen) != Z_OK ||
                    dl != datalen)
                { free(inflated); return -1; }
            }
            content = inflated;
        }
        int16_t version;
        memcpy(&version, content, 2);
        if (version == 0)                       /* Original */
        {
            int16_t oflags, ndatum;
            int32_t nullsz, sz;
            memcpy(&oflags, content + 2, 2);
            memcpy(&ndatum, content + 4, 2);
            memcpy(&nullsz, content + 8, 4);
            memcpy(&sz, content + 12, 4);
            if ((uint32_t) ndatum != rows || row + rows > cap) return -1;
            if (!(oflags & 1))
            {
                if (sz != (int32_t) rows * width || nullsz != 0) return -1;
                memcpy(dst + row * (int64_t) width, content + 16, sz);
                if (out_validity)
                    memset(out_validity + row, 1, rows);
            }
            else
            {
                /* null bitmap (1 bit/row, ON = null) at 16, datums at
                 * 16 + MAXALIGNed bitmap (datumstreamblock.c:3694-3733) */
                const uint8_t *nbmp = content + 16;
                const uint8_t *vals = content + 16 + nullsz;
                int32_t vi = 0;
                if (out_validity == NULL) return -1;
                if (nullsz != (int32_t) ((((rows + 7) >> 3) + 7) & ~7))
                    return -1;
                for (uint32_t r = 0; r < rows; r++)
                {
                    if ((nbmp[r >> 3] >> (r & 7)) & 1)
                    {
                        memset(dst + (row + r) * (int64_t) width, 0, width);
                        out_validity[row + r] = 0;
                    }
                    else
                    {
                        memcpy(dst + (row + r) * (int64_t) width,
                               vals + (int64_t) vi * width, width);
                        out_validity[row + r] = 1;
                        vi++;
                    }
                }
                if (vi * width != sz) return -1;
            }
            row += rows;
        }
        else if (version == 1 || version == 2)  /* Dense / Dense_Enhanced */
        {
            int64_t got = decode_dense_content_v(
                content, width, dst + row * (int64_t) width,
                out_validity ? out_validity + row : NULL, cap - row);
            if (got < 0 || (uint32_t) got != rows) return -1;
            row += got;
        }
        else
        {
            free(inflated);
            return -1;
        }
        free(inflated);
        off += blocklen;
    }
    return row;
}


/* ======================================================================
 * RLE_TYPE codec — DatumStreamVersion_Dense_Enhanced, no-null no-delta
 * subset (compresstype=rle_type, compresslevel=1: no bulk compression,
 * datumstream.c:397-418).  Formats followed:
 *   Dense header      datumstreamblock.h:108-136 (version=2, flags=RLE)
 *   Rle_Extension     datumstreamblock.h:142-172
 *   compress bitmap   LSB-first per byte, one bit per PHYSICAL datum
 *                     (ON = datum carries a repeat count), reader walk
 *                     datumstreamblock.h:1754-1912
 *   repeat counts     Int32Compress varint (1-4 B, top 2 bits = len-1),
 *                     datumstreamblock.h:604-700; count = EXTRA repeats
 *   AO envelope       SmallContent when logical rows ≤ 16383 else
 *                     NonBulkDenseContent (both regular 8-B headers,
 *                     cdbappendonlystorage.h:76-96, _int.h:253-347)
 * Capacity mirrors the writer's reservation: content < blocksize − 32
 * (maxAoHeaderSize for Dense streams, datumstream.c:597-605).
 * The encoder below (dense_encode) is a full restatement of the writer's
 * state machine and is BYTE-EXACT with the compiled reference writer —
 * see the block comment above dense_encode and tests/test_oracle_cpu.py.
 * ====================================================================== */

static int32_t varint_size(int32_t v)
{
    if (v <= 0x3F) return 1;
    if (v <= 0x3FFF) return 2;
    if (v <= 0x3FFFFF) return 3;
    return 4;
}

static int32_t varint_encode(uint8_t *b, int32_t v)
{
    if (v <= 0x3F) { b[0] = (uint8_t) v; return 1; }
    if (v <= 0x3FFF) { b[0] = (1 << 6) | (uint8_t) (v >> 8); b[1] = (uint8_t) v; return 2; }
    if (v <= 0x3FFFFF) { b[0] = (2 << 6) | (uint8_t) (v >> 16); b[1] = (uint8_t) (v >> 8); b[2] = (uint8_t) v; return 3; }
    b[0] = (3 << 6) | (uint8_t) (v >> 24); b[1] = (uint8_t) (v >> 16);
    b[2] = (uint8_t) (v >> 8); b[3] = (uint8_t) v;
    return 4;
}

static int32_t varint_decode(const uint8_t *b, int32_t *len)
{
    int32_t n = (b[0] >> 6) + 1;
    int32_t v = b[0] & 0x3F;
    for (int32_t i = 1; i < n; i++) v = (v << 8) | b[i];
    *len = n;
    return v;
}

/* ======================================================================
 * Dense_Enhanced RLE_TYPE / DELTA writer — EXACT restatement of the
 * reference's DatumStreamBlockWrite state machine for fixed-width,
 * non-null columns (compresstype=rle_type):
 *   Put flow        datumstreamblock.c:3341-3541 (PutDense fixed-width)
 *   capacity        :1992-2330 (DenseHasSpaceItem/Repeat/Delta),
 *                   :1944-1990 (DenseRleSpace, incl. the pending
 *                   repeat-count reservation)
 *   lazy finalize   :2435-2470 (RleFinalizeRepeatCountSize)
 *   repeat marking  :2477-2672 (RleMarkRepeat/RleIncrRepeated, with the
 *                   compress-bitmap zero-fill to phys-1+deltaOnCount)
 *   delta           :2989-3091 (PerformDeltaCompression/DeltaAdd;
 *                   MAX_DELTA 0x1FFFFFFF, widths 4/8 only, unsigned
 *                   compare, delta==0 becomes a DELTA item)
 *   block assembly  :3803-4246 (BlockDense: Dense hdr, Rle_Extension,
 *                   Delta_Extension, bitmaps, varints, MAXALIGN pad)
 *   AO envelope     SmallContent when rows <= 0x3FFF else NonBulkDense
 *                   (cdbappendonlystorage.h:23,30), CRC32C pair
 * maxDataBlockSize = blocksize - 32 (maxAoHeaderSize for Dense streams,
 * datumstream.c:588-605).  Byte-for-byte equality with the COMPILED
 * reference writer (oracle/_ref/libpgwriter.so) is asserted in
 * tests/test_oracle_cpu.py.
 * ====================================================================== */

/* sign+magnitude delta varint (Reserved3): top 2 bits = len-1, bit 5 =
 * POSITIVE flag — datumstreamblock.h:790-930 */
static int32_t varint3_size(int64_t mag)
{
    if (mag <= 0x1F) return 1;
    if (mag <= 0x1FFF) return 2;
    if (mag <= 0x1FFFFF) return 3;
    return 4;
}
static int32_t varint3_encode(uint8_t *b, int64_t mag, int positive)
{
    int32_t n;
    if (mag <= 0x1F) { b[0] = (uint8_t) mag; n = 1; }
    else if (mag <= 0x1FFF) { b[0] = (1 << 6) | (uint8_t) (mag >> 8); b[1] = (uint8_t) mag; n = 2; }
    else if (mag <= 0x1FFFFF) { b[0] = (2 << 6) | (uint8_t) (mag >> 16); b[1] = (uint8_t) (mag >> 8); b[2] = (uint8_t) mag; n = 3; }
    else { b[0] = (3 << 6) | (uint8_t) (mag >> 24); b[1] = (uint8_t) (mag >> 16); b[2] = (uint8_t) (mag >> 8); b[3] = (uint8_t) mag; n = 4; }
    if (positive) b[0] |= 0x20;
    return n;
}
static int64_t varint3_decode(const uint8_t *b, int32_t *len, int *positive)
{
    int32_t n = (b[0] >> 6) + 1;
    *positive = (b[0] >> 5) & 1;
    int64_t v = b[0] & 0x1F;
    for (int32_t i = 1; i < n; i++) v = (v << 8) | b[i];
    *len = n;
    return v;
}

static uint64_t item_at(const uint8_t *v, int width, int64_t i)
{
    uint64_t x = 0;
    memcpy(&x, v + i * width, width);
    return x;
}

/* bit-map writer, LSB-first per byte (datumstreamblock.h:375-424) */
typedef struct {
    uint8_t *buf;
    int32_t bits, on;
} dbm_t;

static void dbm_reset(dbm_t *b) { b->bits = 0; b->on = 0; }
static void dbm_add(dbm_t *b, int on)
{
    if ((b->bits & 7) == 0)
        b->buf[b->bits >> 3] = 0;
    if (on)
    {
        b->buf[b->bits >> 3] |= (uint8_t) (1u << (b->bits & 7));
        b->on++;
    }
    b->bits++;
}
static void dbm_set_last(dbm_t *b)          /* BitMapWrite_Set on current */
{
    b->buf[(b->bits - 1) >> 3] |= (uint8_t) (1u << ((b->bits - 1) & 7));
    b->on++;
}
static void dbm_zerofill(dbm_t *b, int32_t n)   /* :330-368 */
{
    memset(b->buf, 0, (size_t) ((n + 7) >> 3));
    b->bits = n;
    b->on = 0;
}
#define DBM_SIZE(b) (((b)->bits + 7) >> 3)

#define DWR_MAXALIGN(x) (((x) + 7) & ~7)
#define DWR_MAXDATUM 0x3FFFFFFF          /* MAXDATUM_PER_AOCS_DENSE_BLOCK */
#define DWR_MAXREPEAT 0x3FFFFFFF         /* MAXREPEAT_COUNT, datumstreamblock.h:991 */

typedef struct {
    int width, rle_want, delta_want;
    int32_t maxdata;                     /* maxDataBlockSize */
    int32_t nth, phys;                   /* logical rows / physical datums */
    uint8_t *datum_buffer;
    int64_t datum_used;
    /* NULL state */
    int has_null;
    int32_t always;                      /* always_null_bitmap_count */
    dbm_t nbm;                           /* null bitmap */
    /* RLE_TYPE state */
    int rle_has, last_valid, last_repeated;
    uint64_t last_item;
    dbm_t cbm;                           /* compress bitmap */
    int32_t *repeats;
    int32_t nrepeats, repeats_size;      /* repeats_size = FINALIZED runs only */
    /* varlena RLE: stored payload of the last item (offset into
     * datum_buffer) — equality is on PAYLOAD bytes (:3240-3267) */
    int64_t vl_last_off;
    int32_t vl_last_len;
    /* DELTA state */
    int delta_has, not_first;
    uint64_t compare_item;
    dbm_t dbm;                           /* delta bitmap */
    int64_t *deltas;
    uint8_t *dsigns;
    int32_t ndeltas, deltas_size;
} dwr_t;

static void dwr_getready(dwr_t *w)       /* GetReady, :3588-3669 */
{
    w->nth = 0; w->phys = 0; w->datum_used = 0;
    w->has_null = 0; w->always = 0;
    dbm_reset(&w->nbm);
    w->rle_has = 0; w->last_valid = 0; w->last_repeated = 0; w->last_item = 0;
    dbm_reset(&w->cbm);
    w->nrepeats = 0; w->repeats_size = 0;
    w->delta_has = 0; w->not_first = 0; w->compare_item = 0;
    dbm_reset(&w->dbm);
    w->ndeltas = 0; w->deltas_size = 0;
}

/* DenseRleSpace (:1944-1990); for_null uses the CURRENT compress-bitmap
 * size (nulls add no compress bits) */
static void dwr_rle_space(const dwr_t *w, int for_null,
                          int32_t *hdr, int32_t *rle)
{
    if (!w->rle_has)
        return;
    *hdr += 16;                                  /* Rle_Extension */
    *rle += for_null ? (w->cbm.bits + 7) >> 3
                     : (w->cbm.bits + 1 + 7) >> 3;   /* Size vs NextSize */
    *rle += w->repeats_size;
    if (w->last_repeated)                        /* pending finalize reservation */
        *rle += varint_size(w->repeats[w->nrepeats - 1]);
}

static int dwr_has_space_item(const dwr_t *w, int32_t sz)   /* :2330-2378 */
{
    int32_t hdr = 16, rle = 0, delta = 0;
    int32_t nul = w->has_null ? (w->always + 1 + 7) >> 3 : 0;
    if (w->nth + 1 >= DWR_MAXDATUM)
        return 0;
    dwr_rle_space(w, 0, &hdr, &rle);
    if (w->delta_has)
    {
        hdr += 12;                               /* Delta_Extension */
        delta = ((w->dbm.bits + 1 + 7) >> 3) + w->deltas_size;
    }
    return DWR_MAXALIGN(hdr + nul + rle + delta) + w->datum_used + sz <= w->maxdata;
}

static int dwr_has_space_repeat(const dwr_t *w, int new_repeat)  /* :2098-2220 */
{
    int32_t hdr = 16, delta = 0, rle, total;
    int32_t nul = w->has_null ? (w->always + 7) >> 3 : 0;   /* no new bit */
    if (w->nth + 1 >= DWR_MAXDATUM)
        return 0;
    total = w->phys;
    if (w->delta_has)
    {
        hdr += 12;
        delta = ((w->dbm.bits + 1 + 7) >> 3) + w->deltas_size;
        total += w->dbm.on;
    }
    hdr += 16;                                   /* Rle_Extension, unconditional */
    rle = (total + (new_repeat ? 1 : 0) + 7) >> 3;
    rle += w->repeats_size + 4;                  /* + Int32Compress_MaxByteLen */
    return DWR_MAXALIGN(hdr + nul + rle + delta) + w->datum_used <= w->maxdata;
}

static int dwr_has_space_delta(const dwr_t *w)   /* :2224-2328 */
{
    int32_t hdr = 16, rle = 0, total, delta;
    int32_t nul = w->has_null ? (w->always + 1 + 7) >> 3 : 0;
    if (w->nth + 1 >= DWR_MAXDATUM)
        return 0;
    dwr_rle_space(w, 0, &hdr, &rle);
    total = w->phys + (w->delta_has ? w->dbm.on : 0);
    hdr += 12;
    delta = ((total + 1 + 7) >> 3) + w->deltas_size + 4;  /* + Reserved3_MaxByteLen */
    return DWR_MAXALIGN(hdr + nul + rle + delta) + w->datum_used <= w->maxdata;
}

static int dwr_has_space_null(const dwr_t *w)    /* :1992-2090 */
{
    int32_t hdr = 16, rle = 0, delta = 0;
    int32_t nul = (w->always + 1 + 7) >> 3;      /* unconditional */
    if (w->nth + 1 >= DWR_MAXDATUM)
        return 0;
    dwr_rle_space(w, 1, &hdr, &rle);
    if (w->delta_has)
    {
        hdr += 12;
        delta = ((w->dbm.bits + 1 + 7) >> 3) + w->deltas_size;  /* NextSize */
    }
    return DWR_MAXALIGN(hdr + nul + rle + delta) + w->datum_used <= w->maxdata;
}

static void dwr_finalize_repeat(dwr_t *w)        /* :2435-2470 */
{
    w->last_repeated = 0;
    w->last_valid = 0;
    w->repeats_size += varint_size(w->repeats[w->nrepeats - 1]);
}

static void dwr_incr_repeated(dwr_t *w)          /* RleIncrRepeated, :2558-2651 */
{
    if (!w->last_repeated)
    {
        w->last_repeated = 1;
        if (!w->rle_has)
        {
            /* zero-fill a bit per prior physical+delta item, then mark this
             * one repeated (:1880-1886) */
            dbm_zerofill(&w->cbm, w->phys - 1 + (w->delta_has ? w->dbm.on : 0));
            dbm_add(&w->cbm, 1);
            w->rle_has = 1;
        }
        else
            dbm_set_last(&w->cbm);
        w->repeats[w->nrepeats++] = 1;           /* count = EXTRA repeats */
    }
    else
        w->repeats[w->nrepeats - 1]++;
    w->nth++;
}

/* PerformDeltaCompression (:2989-3091): 0 = OK, 1 = ERROR, 2 = NOT_APPLIED */
static int dwr_perform_delta(dwr_t *w, uint64_t v)
{
    uint64_t mag;
    int positive;
    if (!w->delta_want)
        return 2;
    if (!w->not_first)                   /* first datum of block stays physical */
    {
        w->not_first = 1;
        return 2;
    }
    if (w->width == 4)
    {
        uint32_t c = (uint32_t) w->compare_item, d = (uint32_t) v;
        if (c <= d) { mag = d - c; positive = 1; }
        else        { mag = c - d; positive = 0; }
    }
    else
    {
        if (w->compare_item <= v) { mag = v - w->compare_item; positive = 1; }
        else                      { mag = w->compare_item - v; positive = 0; }
        if (mag > 0x7FFFFFFFFFFFFFFFULL)         /* int64 overflow → delta < 0 */
            return 2;
    }
    if (mag > 0x1FFFFFFF)                        /* MAX_DELTA_SUPPORTED... */
        return 2;
    if (!dwr_has_space_delta(w))
        return 1;
    w->compare_item = v;
    /* DeltaAdd (:2847-2987) */
    if (!w->delta_has)
        dbm_zerofill(&w->dbm, w->phys);
    w->delta_has = 1;
    if (w->has_null)
        dbm_add(&w->nbm, 0);
    w->always++;
    if (w->last_repeated)
        dwr_finalize_repeat(w);
    w->last_item = v;
    w->last_valid = 1;
    if (w->rle_has)
        dbm_add(&w->cbm, 0);
    dbm_add(&w->dbm, 1);
    w->deltas[w->ndeltas] = (int64_t) mag;
    w->dsigns[w->ndeltas] = (uint8_t) positive;
    w->deltas_size += varint3_size((int64_t) mag);
    w->ndeltas++;
    w->nth++;
    return 0;
}

/* PutDense, fixed-width (:3094-3541); >0 stored, 0 folded/null, <0 full */
static int dwr_put(dwr_t *w, uint64_t v, int isnull)
{
    int have_prev;
    if (isnull)
    {
        if (!dwr_has_space_null(w))
            return -1;
        if (!w->has_null)                /* MakeNullBitMapSpace first-null */
        {
            w->has_null = 1;
            dbm_zerofill(&w->nbm, w->always);
        }
        dbm_add(&w->nbm, 1);             /* DenseIncrNull */
        w->always++;
        if (w->rle_want)
        {
            if (w->last_repeated)
                dwr_finalize_repeat(w);
            w->last_valid = 0;
        }
        w->nth++;
        return 0;
    }
    have_prev = w->rle_want && w->last_valid;
    if (w->last_repeated && w->repeats[w->nrepeats - 1] >= DWR_MAXREPEAT)
        dwr_finalize_repeat(w);
    else if (have_prev)
    {
        int eq = (w->width == 8) ? (w->last_item == v)
               : (w->width == 4) ? ((uint32_t) w->last_item == (uint32_t) v)
               : (w->width == 2) ? ((uint16_t) w->last_item == (uint16_t) v)
               : ((uint8_t) w->last_item == (uint8_t) v);
        if (eq)
        {
            if (!dwr_has_space_repeat(w, !w->last_repeated))
                return -1;
            dwr_incr_repeated(w);
            return 0;
        }
        if (w->last_repeated)
            dwr_finalize_repeat(w);
    }
    if (w->delta_want)
    {
        int st = dwr_perform_delta(w, v);
        if (st == 0) return 0;
        if (st == 1) return -1;
    }
    if (!dwr_has_space_item(w, w->width))
        return -w->width;
    memcpy(w->datum_buffer + w->datum_used, &v, w->width);
    w->datum_used += w->width;
    /* DenseIncrItem (:2506-2556) */
    if (w->has_null)
        dbm_add(&w->nbm, 0);
    w->always++;
    if (w->last_repeated)
        dwr_finalize_repeat(w);
    if (w->rle_want)
    {
        w->last_item = v;
        w->last_valid = 1;
    }
    if (w->rle_has)
        dbm_add(&w->cbm, 0);
    w->nth++;
    w->phys++;
    /* DeltaMaintain (:2800-2845) */
    if (w->delta_want)
    {
        w->compare_item = v;
        if (w->delta_has)
            dbm_add(&w->dbm, 0);
    }
    return w->width;
}

/* BlockDense (:3803-4246) + AO envelope; returns whole-block length */
static int64_t dwr_block(dwr_t *w, int64_t first_rownum,
                         uint8_t *blk, int64_t cap)
{
    int32_t hdr, rle, delta, meta, aligned;
    int64_t content, blocklen;
    int16_t v16;
    int32_t v32;
    uint8_t *c, *p;
    uint32_t kind, b03, b47;

    if (w->last_repeated)
        dwr_finalize_repeat(w);
    hdr = 16 + (w->rle_has ? 16 : 0) + (w->delta_has ? 12 : 0);
    rle = w->rle_has ? DBM_SIZE(&w->cbm) + w->repeats_size : 0;
    delta = w->delta_has ? DBM_SIZE(&w->dbm) + w->deltas_size : 0;
    meta = hdr + (w->has_null ? DBM_SIZE(&w->nbm) : 0) + rle + delta;
    aligned = DWR_MAXALIGN(meta);
    content = aligned + w->datum_used;
    blocklen = (24 + content + 7) & ~7LL;
    if (blocklen > cap)
        return -1;
    memset(blk, 0, (size_t) blocklen);

    kind = (w->nth <= 0x3FFF) ? 1u : 3u;         /* Small / NonBulkDense */
    b03 = (kind << 28) | (1u << 27) | (1u << 24);
    b47 = 0;
    if (kind == 1)
    {
        b03 |= (0x00FFFC00u & ((uint32_t) w->nth << 10)) |
               (((uint32_t) content >> 11) & 0x3FFu);
        b47 = (((uint32_t) content & 0x7FFu) << 21);
    }
    else
    {
        b03 |= ((uint32_t) content & 0x1FFFFFu);
        b47 = (uint32_t) w->nth & 0x3FFFFFFFu;
    }
    put_u32le(blk, b03);
    put_u32le(blk + 4, b47);
    memcpy(blk + 16, &first_rownum, 8);

    c = blk + 24;
    v16 = 2;            memcpy(c, &v16, 2);      /* Dense_Enhanced */
    v16 = (int16_t) ((w->has_null ? 1 : 0) | (w->rle_has ? 2 : 0) |
                     (w->delta_has ? 4 : 0));
    memcpy(c + 2, &v16, 2);
    v32 = w->nth;       memcpy(c + 4, &v32, 4);
    v32 = w->phys;      memcpy(c + 8, &v32, 4);
    v32 = (int32_t) w->datum_used; memcpy(c + 12, &v32, 4);
    p = c + 16;
    if (w->rle_has)
    {
        v32 = w->has_null ? w->nbm.bits : 0;            /* norepeats count */
        memcpy(p, &v32, 4);
        v32 = w->cbm.bits;     memcpy(p + 4, &v32, 4);
        v32 = w->nrepeats;     memcpy(p + 8, &v32, 4);
        v32 = w->repeats_size; memcpy(p + 12, &v32, 4);
        p += 16;
    }
    if (w->delta_has)
    {
        v32 = w->dbm.bits;    memcpy(p, &v32, 4);
        v32 = w->ndeltas;     memcpy(p + 4, &v32, 4);
        v32 = w->deltas_size; memcpy(p + 8, &v32, 4);
        p += 12;
    }
    if (w->has_null)
    {
        memcpy(p, w->nbm.buf, DBM_SIZE(&w->nbm));
        p += DBM_SIZE(&w->nbm);
    }
    if (w->rle_has)
    {
        memcpy(p, w->cbm.buf, DBM_SIZE(&w->cbm));
        p += DBM_SIZE(&w->cbm);
        for (int32_t i = 0; i < w->nrepeats; i++)
            p += varint_encode(p, w->repeats[i]);
    }
    if (w->delta_has)
    {
        memcpy(p, w->dbm.buf, DBM_SIZE(&w->dbm));
        p += DBM_SIZE(&w->dbm);
        for (int32_t i = 0; i < w->ndeltas; i++)
            p += varint3_encode(p, w->deltas[i], w->dsigns[i]);
    }
    memcpy(blk + 24 + aligned, w->datum_buffer, (size_t) w->datum_used);

    put_u32le(blk + 8, orc_crc32c(0xFFFFFFFFu, blk + 16, blocklen - 16));
    put_u32le(blk + 12, orc_crc32c(0xFFFFFFFFu, blk, 12));
    return blocklen;
}

static int64_t dense_encode(const void *vals, const uint8_t *nulls,
                            int width, int64_t nrows,
                            int64_t first_rownum, int32_t blocksize,
                            int rle_want, int delta_want,
                            uint8_t *out, int64_t outcap)
{
    const uint8_t *src = (const uint8_t *) vals;
    dwr_t w;
    int64_t off = 0, emitted = 0;
    size_t scratch;

    if (width != 1 && width != 2 && width != 4 && width != 8)
        return -1;
    if (delta_want && width != 4 && width != 8)
        return -1;                       /* DeltaMaintain FATALs otherwise */
    memset(&w, 0, sizeof(w));
    w.width = width;
    w.rle_want = rle_want;
    w.delta_want = delta_want;
    w.maxdata = blocksize - 32;          /* Dense maxAoHeaderSize reserve */
    scratch = (size_t) blocksize * 2 + 64;
    w.datum_buffer = malloc((size_t) w.maxdata + 16);
    w.cbm.buf = malloc(scratch);
    w.dbm.buf = malloc(scratch);
    w.nbm.buf = malloc(scratch);
    w.repeats = malloc(scratch * sizeof(int32_t));
    w.deltas = malloc(scratch * sizeof(int64_t));
    w.dsigns = malloc(scratch);
    if (!w.datum_buffer || !w.cbm.buf || !w.dbm.buf || !w.nbm.buf ||
        !w.repeats || !w.deltas || !w.dsigns)
        goto fail;
    dwr_getready(&w);
    for (int64_t i = 0; i < nrows; i++)
    {
        int isnull = nulls != NULL && nulls[i] != 0;
        uint64_t v = isnull ? 0 : item_at(src, width, i);
        if (dwr_put(&w, v, isnull) < 0)
        {
            int64_t bl = dwr_block(&w, first_rownum + emitted,
                                   out + off, outcap - off);
            if (bl < 0)
                goto fail;
            off += bl;
            emitted += w.nth;
            dwr_getready(&w);
            if (dwr_put(&w, v, isnull) < 0)
                goto fail;
        }
    }
    if (w.nth > 0)
    {
        int64_t bl = dwr_block(&w, first_rownum + emitted,
                               out + off, outcap - off);
        if (bl < 0)
            goto fail;
        off += bl;
    }
    free(w.datum_buffer); free(w.cbm.buf); free(w.dbm.buf); free(w.nbm.buf);
    free(w.repeats); free(w.deltas); free(w.dsigns);
    return off;
fail:
    free(w.datum_buffer); free(w.cbm.buf); free(w.dbm.buf); free(w.nbm.buf);
    free(w.repeats); free(w.deltas); free(w.dsigns);
    return -1;
}

int64_t orc_aocs_encode_rle(const void *vals, int width, int64_t nrows,
                            int64_t first_rownum, int32_t blocksize,
                            uint8_t *out, int64_t outcap)
{
    return dense_encode(vals, NULL, width, nrows, first_rownum, blocksize,
                        1, 0, out, outcap);
}

int64_t orc_aocs_encode_rle_delta(const void *vals, int width, int64_t nrows,
                                  int64_t first_rownum, int32_t blocksize,
                                  uint8_t *out, int64_t outcap)
{
    return dense_encode(vals, NULL, width, nrows, first_rownum, blocksize,
                        1, 1, out, outcap);
}

int64_t orc_aocs_encode_rle_delta_nulls(const void *vals, const uint8_t *nulls,
                                        int width, int64_t nrows,
                                        int64_t first_rownum, int32_t blocksize,
                                        int delta,
                                        uint8_t *out, int64_t outcap)
{
    return dense_encode(vals, nulls, width, nrows, first_rownum, blocksize,
                        1, delta, out, outcap);
}

/* Dense_Enhanced varlena put (:3160-3380): RLE equality on payload bytes,
 * short-form / aligned 4-byte storage as in Orig, same capacity machinery
 * as the fixed-width dense writer; no delta for varlena. */
static int dwr_put_varlena(dwr_t *w, const uint8_t *payload, int64_t len,
                           int isnull)
{
    int have_prev;
    if (isnull)
    {
        if (!dwr_has_space_null(w))
            return -1;
        if (!w->has_null)
        {
            w->has_null = 1;
            dbm_zerofill(&w->nbm, w->always);
        }
        dbm_add(&w->nbm, 1);
        w->always++;
        if (w->rle_want)
        {
            if (w->last_repeated)
                dwr_finalize_repeat(w);
            w->last_valid = 0;
        }
        w->nth++;
        return 0;
    }
    have_prev = w->rle_want && w->last_valid;
    if (w->last_repeated && w->repeats[w->nrepeats - 1] >= DWR_MAXREPEAT)
        dwr_finalize_repeat(w);
    else if (have_prev)
    {
        int eq = (int64_t) w->vl_last_len == len &&
                 memcmp(w->datum_buffer + w->vl_last_off, payload,
                        (size_t) len) == 0;
        if (eq)
        {
            if (!dwr_has_space_repeat(w, !w->last_repeated))
                return -1;
            dwr_incr_repeated(w);
            return 0;
        }
        if (w->last_repeated)
            dwr_finalize_repeat(w);
    }
    int64_t sz, pad = 0;
    if (len + 1 <= 0x7F)
        sz = len + 1;
    else
    {
        pad = (-w->datum_used) & 3;          /* att_align_zero, pad stays
                                                even if the check fails */
        sz = len + 4;
    }
    if (pad)
    {
        memset(w->datum_buffer + w->datum_used, 0, (size_t) pad);
        w->datum_used += pad;
    }
    if (!dwr_has_space_item(w, (int32_t) sz))
        return (int) -sz;
    int64_t data_off;
    if (len + 1 <= 0x7F)
    {
        w->datum_buffer[w->datum_used] = (uint8_t) (((len + 1) << 1) | 1);
        memcpy(w->datum_buffer + w->datum_used + 1, payload, (size_t) len);
        data_off = w->datum_used + 1;
    }
    else
    {
        uint32_t hdr = (uint32_t) ((len + 4) << 2);
        memcpy(w->datum_buffer + w->datum_used, &hdr, 4);
        memcpy(w->datum_buffer + w->datum_used + 4, payload, (size_t) len);
        data_off = w->datum_used + 4;
    }
    w->datum_used += sz;
    /* DenseIncrItem */
    if (w->has_null)
        dbm_add(&w->nbm, 0);
    w->always++;
    if (w->last_repeated)
        dwr_finalize_repeat(w);
    if (w->rle_want)
    {
        w->vl_last_off = data_off;
        w->vl_last_len = (int32_t) len;
        w->last_valid = 1;
    }
    if (w->rle_has)
        dbm_add(&w->cbm, 0);
    w->nth++;
    w->phys++;
    return (int) sz;
}

int64_t orc_aocs_encode_varlena_rle(const uint8_t *payload,
                                    const int64_t *offsets,
                                    const uint8_t *nulls, int64_t nrows,
                                    int64_t first_rownum, int32_t blocksize,
                                    uint8_t *out, int64_t outcap)
{
    dwr_t w;
    int64_t off = 0, emitted = 0;
    size_t scratch;

    memset(&w, 0, sizeof(w));
    w.width = -1;
    w.rle_want = 1;
    w.delta_want = 0;
    w.maxdata = blocksize - 32;
    scratch = (size_t) blocksize * 2 + 64;
    w.datum_buffer = malloc((size_t) w.maxdata + 16);
    w.cbm.buf = malloc(scratch);
    w.dbm.buf = malloc(scratch);
    w.nbm.buf = malloc(scratch);
    w.repeats = malloc(scratch * sizeof(int32_t));
    w.deltas = malloc(scratch * sizeof(int64_t));
    w.dsigns = malloc(scratch);
    if (!w.datum_buffer || !w.cbm.buf || !w.dbm.buf || !w.nbm.buf ||
        !w.repeats || !w.deltas || !w.dsigns)
        goto fail;
    dwr_getready(&w);
    for (int64_t i = 0; i < nrows; i++)
    {
        int isnull = nulls != NULL && nulls[i] != 0;
        const uint8_t *p = payload + offsets[i];
        int64_t len = offsets[i + 1] - offsets[i];
        if (dwr_put_varlena(&w, p, isnull ? 0 : len, isnull) < 0)
        {
            int64_t bl = dwr_block(&w, first_rownum + emitted,
                                   out + off, outcap - off);
            if (bl < 0)
                goto fail;
            off += bl;
            emitted += w.nth;
            dwr_getready(&w);
            if (dwr_put_varlena(&w, p, isnull ? 0 : len, isnull) < 0)
                goto fail;
        }
    }
    if (w.nth > 0)
    {
        int64_t bl = dwr_block(&w, first_rownum + emitted,
                               out + off, outcap - off);
        if (bl < 0)
            goto fail;
        off += bl;
    }
    free(w.datum_buffer); free(w.cbm.buf); free(w.dbm.buf); free(w.nbm.buf);
    free(w.repeats); free(w.deltas); free(w.dsigns);
    return off;
fail:
    free(w.datum_buffer); free(w.cbm.buf); free(w.dbm.buf); free(w.nbm.buf);
    free(w.repeats); free(w.deltas); free(w.dsigns);
    return -1;
}

/* Original-version writer with NULL support — PutOrig (:1569-1770) +
 * BlockOrig (:3669-3801): header {version,flags,ndatum,enc,nullsz,sz},
 * null bitmap (1 bit/row, ON = null) MAXALIGNed, then packed datums.
 * Capacity: OrigHasSpace (:1508-1566), STRICT '<' and
 * nullSize = MAXALIGN(Size(always+1)) once any null exists. */
int64_t orc_aocs_encode_orig_nulls(const void *vals, const uint8_t *nulls,
                                   int width, int64_t nrows,
                                   int64_t first_rownum, int32_t blocksize,
                                   uint8_t *out, int64_t outcap)
{
    const uint8_t *src = (const uint8_t *) vals;
    int32_t maxdata = blocksize - 24;
    int64_t off = 0, row = 0;
    int32_t cap_rows = blocksize;                /* nth < 16383 anyway */
    uint8_t *dvals = malloc((size_t) maxdata + 16);
    uint8_t *nbm = malloc((size_t) ((cap_rows + 7) >> 3) + 8);
    if (!dvals || !nbm) { free(dvals); free(nbm); return -1; }

    while (row < nrows)
    {
        int32_t nth = 0, phys = 0, always = 0;
        int has_null = 0;
        int64_t used = 0;
        memset(nbm, 0, (size_t) ((cap_rows + 7) >> 3) + 8);
        while (row + nth < nrows && nth + 1 < 16383)
        {
            int isnull = nulls != NULL && nulls[row + nth] != 0;
            int32_t nullsize =
                (isnull || has_null)
                    ? (int32_t) ((((always + 1 + 7) >> 3) + 7) & ~7) : 0;
            int32_t sz = isnull ? 0 : width;
            if (!(16 + nullsize + used + sz < maxdata))
                break;
            if (isnull)
            {
                has_null = 1;
                nbm[always >> 3] |= (uint8_t) (1u << (always & 7));
            }
            else
            {
                memcpy(dvals + used, src + (row + nth) * width, width);
                used += width;
                phys++;
            }
            always++;
            nth++;
        }
        if (nth == 0) { free(dvals); free(nbm); return -1; }

        int32_t nullsz = has_null
            ? (int32_t) ((((nth + 7) >> 3) + 7) & ~7) : 0;
        int32_t content = 16 + nullsz + (int32_t) used;
        int64_t blocklen = (24 + content + 7) & ~7LL;
        if (off + blocklen > outcap) { free(dvals); free(nbm); return -1; }
        uint8_t *blk = out + off;
        memset(blk, 0, (size_t) blocklen);
        uint32_t b03 = (1u << 28) | (1u << 27) | (1u << 24) |
                       (0x00FFFC00u & ((uint32_t) nth << 10)) |
                       (((uint32_t) content >> 11) & 0x3FFu);
        uint32_t b47 = (((uint32_t) content & 0x7FFu) << 21);
        put_u32le(blk, b03);
        put_u32le(blk + 4, b47);
        int64_t frn = first_rownum + row;
        memcpy(blk + 16, &frn, 8);
        uint8_t *c = blk + 24;
        int16_t v16 = 0;  memcpy(c, &v16, 2);
        v16 = has_null ? 1 : 0; memcpy(c + 2, &v16, 2);
        v16 = (int16_t) nth; memcpy(c + 4, &v16, 2);
        v16 = 0; memcpy(c + 6, &v16, 2);
        int32_t v32 = nullsz; memcpy(c + 8, &v32, 4);
        v32 = (int32_t) used; memcpy(c + 12, &v32, 4);
        if (has_null)
            memcpy(c + 16, nbm, (size_t) ((nth + 7) >> 3));
        memcpy(c + 16 + nullsz, dvals, (size_t) used);
        put_u32le(blk + 8, orc_crc32c(0xFFFFFFFFu, blk + 16, blocklen - 16));
        put_u32le(blk + 12, orc_crc32c(0xFFFFFFFFu, blk, 12));
        off += blocklen;
        row += nth;
    }
    free(dvals); free(nbm);
    return off;
}



static int64_t encode_bulk(const void *vals, int width, int64_t nrows,
                           int64_t first_rownum, int32_t blocksize,
                           int level, int codec, uint8_t *out, int64_t outcap)
{
    int32_t rpb = orc_aocs_rows_per_block(width, blocksize);
    const uint8_t *src = (const uint8_t *) vals;
    uint8_t *content = malloc(blocksize + 16);
    uint8_t *comp = malloc(compressBound(blocksize) + blocksize + 512);
    int64_t off = 0, row = 0;
    while (row < nrows)
    {
        int32_t rows = (int32_t) ((nrows - row < rpb) ? (nrows - row) : rpb);
        int32_t sz = rows * width;
        int32_t clen = 16 + sz;                 /* Orig content */
        int16_t v16;
        int32_t v32;
        v16 = 0;              memcpy(content + 0, &v16, 2);
        v16 = 0;              memcpy(content + 2, &v16, 2);
        v16 = (int16_t) rows; memcpy(content + 4, &v16, 2);
        v16 = 0;              memcpy(content + 6, &v16, 2);
        v32 = 0;              memcpy(content + 8, &v32, 4);
        v32 = sz;             memcpy(content + 12, &v32, 4);
        memcpy(content + 16, src + row * (int64_t) width, sz);

        unsigned long dlen;
        int use_comp;
        if (codec == 2)
        {
            size_t r = ZSTD_compress(comp, ZSTD_compressBound(clen),
                                     content, clen, level);
            use_comp = !ZSTD_isError(r) && (int64_t) r < clen;
            dlen = (unsigned long) r;
        }
        else
        {
            dlen = compressBound(clen);
            int zrc = compress2(comp, &dlen, content, clen, level);
            use_comp = (zrc == Z_OK) && ((int64_t) dlen < clen);
        }
        int32_t stored = use_comp ? (int32_t) dlen : clen;
        const uint8_t *body = use_comp ? comp : content;
        int64_t blocklen = (24 + stored + 7) & ~7LL;
        if (off + blocklen > outcap) { free(content); free(comp); return -1; }
        uint8_t *blk = out + off;
        memset(blk, 0, blocklen);
        uint32_t b03 = (1u << 28) | (1u << 27) | (1u << 24) |
                       (0x00FFFC00u & ((uint32_t) rows << 10)) |
                       (((uint32_t) clen >> 11) & 0x3FFu);
        uint32_t b47 = (((uint32_t) clen & 0x7FFu) << 21) |
                       (use_comp ? ((uint32_t) stored & 0x1FFFFFu) : 0u);
        put_u32le(blk, b03);
        put_u32le(blk + 4, b47);
        int64_t frn = first_rownum + row;
        memcpy(blk + 16, &frn, 8);
        memcpy(blk + 24, body, stored);
        put_u32le(blk + 8, orc_crc32c(0xFFFFFFFFu, blk + 16, blocklen - 16));
        put_u32le(blk + 12, orc_crc32c(0xFFFFFFFFu, blk, 12));
        off += blocklen;
        row += rows;
    }
    free(content); free(comp);
    return off;
}

int64_t orc_aocs_encode_zlib(const void *vals, int width, int64_t nrows,
                             int64_t first_rownum, int32_t blocksize,
                             int level, uint8_t *out, int64_t outcap)
{
    return encode_bulk(vals, width, nrows, first_rownum, blocksize, level, 1,
                       out, outcap);
}

int64_t orc_aocs_encode_zstd(const void *vals, int width, int64_t nrows,
                             int64_t first_rownum, int32_t blocksize,
                             int level, uint8_t *out, int64_t outcap)
{
    return encode_bulk(vals, width, nrows, first_rownum, blocksize, level, 2,
                       out, outcap);
}

/* decode one Dense(±RLE±DELTA) content area; returns rows written or -1.
 * Walker follows DatumStreamBlockRead_AdvanceDense/…DenseDelta
 * (datumstreamblock.h:1624-1912): per NEW item advance the compress bitmap
 * (repeat count varint when ON) and the delta bitmap (signed-magnitude
 * varint applied to the running value when ON; physical datum otherwise). */
/* Dense/Dense_Enhanced content walker, full feature set: RLE_TYPE,
 * DELTA_RANGE and the NULL bitmap (one bit per NON-REPEAT logical slot,
 * ON = null; reader walk datumstreamblock.h:1754-1912, section order
 * datumstreamblock.c:3963-4040).  validity gets one byte per row
 * (1 = non-null); NULL validity refuses null-bearing blocks. */
static int64_t decode_dense_content_v(const uint8_t *c, int width,
                                      uint8_t *dst, uint8_t *validity,
                                      int64_t cap_rows)
{
    int16_t version, flags;
    int32_t logical, phys, psize;
    memcpy(&version, c, 2);
    memcpy(&flags, c + 2, 2);
    memcpy(&logical, c + 4, 4);
    memcpy(&phys, c + 8, 4);
    memcpy(&psize, c + 12, 4);
    if (psize != phys * width || logical > cap_rows) return -1;
    int has_null = (flags & 0x1) != 0;
    int rle = (flags & 0x2) != 0, delta = (flags & 0x4) != 0;
    if (has_null && validity == NULL) return -1;
    if (!rle && !delta && !has_null)
    {
        if (logical != phys) return -1;
        memcpy(dst, c + 16, (size_t) psize);
        if (validity) memset(validity, 1, (size_t) logical);
        return logical;
    }
    const uint8_t *p = c + 16;
    int32_t bmbits = 0, ncnt = 0, csize = 0;
    int32_t dbmbits = 0, ndelta = 0, dsize = 0;
    int32_t nullbits = has_null ? logical : 0;   /* no-RLE: 1 bit per row */
    if (rle)
    {
        int32_t norepeats;
        memcpy(&norepeats, p, 4);
        memcpy(&bmbits, p + 4, 4);
        memcpy(&ncnt, p + 8, 4);
        memcpy(&csize, p + 12, 4);
        if (has_null)
            nullbits = norepeats;
        else if (norepeats != 0)
            return -1;
        p += 16;
    }
    if (delta)
    {
        memcpy(&dbmbits, p, 4);
        memcpy(&ndelta, p + 4, 4);
        memcpy(&dsize, p + 8, 4);
        p += 12;
    }
    const uint8_t *nbmp = NULL, *bmp = NULL, *cnts = NULL, *dbm = NULL, *dbs = NULL;
    if (has_null)
    {
        nbmp = p;
        p += (nullbits + 7) >> 3;
    }
    if (rle)
    {
        bmp = p; p += (bmbits + 7) >> 3;
        cnts = p; p += csize;
    }
    if (delta)
    {
        dbm = p; p += (dbmbits + 7) >> 3;
        dbs = p; p += dsize;
    }
    int32_t hdr = (int32_t) (p - c);
    const uint8_t *datum = c + ((hdr + 7) & ~7);

    int64_t w = 0;
    int32_t item = 0, phys_idx = 0, coff = 0, doff = 0, dseen = 0, npos = 0;
    uint64_t cur = 0;
    while (w < logical)
    {
        if (has_null)
        {
            if (npos >= nullbits) return -1;
            int nbit = (nbmp[npos >> 3] >> (npos & 7)) & 1;
            npos++;
            if (nbit)
            {
                memset(dst + w * width, 0, (size_t) width);
                validity[w] = 0;
                w++;
                continue;
            }
        }
        if ((rle && item >= bmbits) || (delta && item >= dbmbits)) return -1;
        int64_t reps = 1;
        if (rle && (bmp[item >> 3] & (1u << (item & 7))))
        {
            int32_t len, v = varint_decode(cnts + coff, &len);
            coff += len;
            reps += v;
        }
        if (delta && (dbm[item >> 3] & (1u << (item & 7))))
        {
            int32_t len, pos;
            int64_t mag = varint3_decode(dbs + doff, &len, &pos);
            doff += len;
            dseen++;
            if (width == 8)
                cur = pos ? cur + (uint64_t) mag : cur - (uint64_t) mag;
            else
                cur = (uint32_t) (pos ? (uint32_t) cur + (uint32_t) mag
                                      : (uint32_t) cur - (uint32_t) mag);
        }
        else
        {
            if (phys_idx >= phys) return -1;
            cur = item_at(datum, width, phys_idx);
            phys_idx++;
        }
        if (w + reps > logical) return -1;
        for (int64_t r = 0; r < reps; r++)
            memcpy(dst + (w + r) * width, &cur, width);
        if (validity)
            memset(validity + w, 1, (size_t) reps);
        w += reps;
        item++;
    }
    if (phys_idx != phys || (rle && (coff != csize || item != bmbits)) ||
        (delta && (doff != dsize || dseen != ndelta || item != dbmbits)) ||
        (has_null && npos != nullbits))
        return -1;
    return w;
}

static int64_t decode_dense_content(const uint8_t *c, int width,
                                    uint8_t *dst, int64_t cap_rows)
{
    return decode_dense_content_v(c, width, dst, NULL, cap_rows);
}

/* ======================================================================
 * Q3 pipeline — reference executor semantics:
 *   scan+filter  execScan.c:161-263 (strict qual; data is NOT NULL)
 *   hash join    nodeHashjoin.c:252-834 (inner, build then probe;
 *                build-before-outer also satisfies the prefetch_inner
 *                Motion-deadlock rule nodeHashjoin.c:332-337)
 *   hash agg     nodeAgg.c:2743,2288 with SUM(float8)=float8pl float.c:769;
 *                group key (l_orderkey,o_orderdate,o_shippriority) is
 *                functionally determined by l_orderkey here.
 * The internal hash-table layout/hash is NOT the reference's (simplehash):
 * parity-irrelevant per SURVEY §8a (result set identical).
 * ====================================================================== */

typedef struct {
    int64_t *keys;               /* EMPTY = INT64_MIN */
    uint64_t mask;
} set64;

static uint64_t hmix64(uint64_t x)   /* internal table hash (not parity-relevant) */
{
    x ^= x >> 33; x *= 0xFF51AFD7ED558CCDULL;
    x ^= x >> 33; x *= 0xC4CEB9FE1A85EC53ULL;
    x ^= x >> 33; return x;
}

#define EMPTY_KEY INT64_MIN

static void set_init(set64 *s, int64_t want)
{
    uint64_t sz = 16;
    while (sz < (uint64_t) want * 2) sz <<= 1;
    s->keys = malloc(sizeof(int64_t) * sz);
    s->mask = sz - 1;
    for (uint64_t i = 0; i < sz; i++) s->keys[i] = EMPTY_KEY;
}
static void set_insert(set64 *s, int64_t k)
{
    uint64_t i = hmix64((uint64_t) k) & s->mask;
    while (1)
    {
        int64_t expect = EMPTY_KEY;
        int64_t cur = __atomic_load_n(&s->keys[i], __ATOMIC_RELAXED);
        if (cur == k) return;
        if (cur == EMPTY_KEY &&
            __atomic_compare_exchange_n(&s->keys[i], &expect, k, 0,
                                        __ATOMIC_ACQ_REL, __ATOMIC_ACQUIRE))
            return;
        if (__atomic_load_n(&s->keys[i], __ATOMIC_RELAXED) == k) return;
        i = (i + 1) & s->mask;
    }
}
static int set_contains(const set64 *s, int64_t k)
{
    uint64_t i = hmix64((uint64_t) k) & s->mask;
    while (s->keys[i] != EMPTY_KEY) { if (s->keys[i] == k) return 1; i = (i + 1) & s->mask; }
    return 0;
}

typedef struct {
    int64_t *key;                /* o_orderkey, EMPTY_KEY empty */
    int32_t *odate;
    int32_t *oprio;
    double  *rev;
    int64_t *cnt;
    uint64_t mask;
} q3tab;

static void tab_init(q3tab *t, int64_t want)
{
    uint64_t sz = 16;
    while (sz < (uint64_t) want * 2) sz <<= 1;
    t->key = malloc(sizeof(int64_t) * sz);
    t->odate = malloc(sizeof(int32_t) * sz);
    t->oprio = malloc(sizeof(int32_t) * sz);
    t->rev = calloc(sz, sizeof(double));
    t->cnt = calloc(sz, sizeof(int64_t));
    t->mask = sz - 1;
    for (uint64_t i = 0; i < sz; i++) t->key[i] = EMPTY_KEY;
}


/* ======================================================================
 * TPC-H Q1 core (BASELINE config 4): GROUP BY l_returnflag, l_linestatus
 * over WHERE l_shipdate <= cutoff — 6 fixed groups with COUNT(*),
 * SUM(l_extendedprice), SUM(l_extendedprice*(1-l_discount)) and
 * AVG = SUM/COUNT (PG float8_avg = Sx/N from the same running sum,
 * float.c:3045 transition, utils/adt/float.c float8_avg final).
 * Streaming-order accumulation → GPU parity at 1e-6 relative.
 * ====================================================================== */
int orc_q1(double sf, uint64_t seed, int32_t cutoff, orc_q1_group *out6)
{
    int64_t nord = orc_norders(sf);
    for (int g = 0; g < 6; g++)
    {
        out6[g].returnflag = (int8_t) (g / 2);
        out6[g].linestatus = (int8_t) (g % 2);
        out6[g].count = 0;
        out6[g].sum_price = 0.0;
        out6[g].sum_revenue = 0.0;
    }
    for (int64_t o = 1; o <= nord; o++)
    {
        int32_t nl = gen_nlines(seed, o);
        for (int32_t j = 0; j < nl; j++)
        {
            if (!(gen_shipdate(seed, o, j) <= cutoff)) continue;
            int g = gen_returnflag(seed, o, j) * 2 + gen_linestatus(seed, o, j);
            double p = gen_price(seed, o, j);
            out6[g].count++;
            out6[g].sum_price += p;
            out6[g].sum_revenue += p * (1.0 - gen_discount(seed, o, j));
        }
    }
    return 0;
}

static int q3n_group_cmp(const void *a, const void *b)
{
    const orc_q3n_group *x = a, *y = b;
    return (x->l_orderkey > y->l_orderkey) - (x->l_orderkey < y->l_orderkey);
}

/*
 * numeric(15,2) Q3 — the real TPC-H column types (tpch500GB.sql:69-72)
 * mapped to scaled int64.  The synthetic f64 measures are
 * exact 2-decimal values by construction (gen_price/gen_discount), so the
 * cents mapping is recovered exactly with llround.  revenue numerator =
 * Σ price_c·(100−disc_c) — bit-exact regardless of ordering, matching the
 * PG numeric SUM for these ranges (no rounding anywhere).
 */
int64_t orc_q3_numeric(const orc_customer *c, const orc_orders *o,
                       const orc_lineitem *l, int32_t cutoff,
                       orc_q3n_group **out)
{
    set64 cust;
    set_init(&cust, c->n + 16);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (int64_t i = 0; i < c->n; i++)
        if (c->c_mktsegment[i] == 0)
            set_insert(&cust, c->c_custkey[i]);

    q3tab t;
    tab_init(&t, o->n + 16);
    int64_t *num = calloc(t.mask + 1, sizeof(int64_t));
    for (int64_t i = 0; i < o->n; i++)
    {
        if (!(o->o_orderdate[i] < cutoff)) continue;
        if (!set_contains(&cust, o->o_custkey[i])) continue;
        int64_t k = o->o_orderkey[i];
        uint64_t j = hmix64((uint64_t) k) & t.mask;
        while (t.key[j] != EMPTY_KEY && t.key[j] != k) j = (j + 1) & t.mask;
        t.key[j] = k;
        t.odate[j] = o->o_orderdate[i];
        t.oprio[j] = o->o_shippriority[i];
    }
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (int64_t i = 0; i < l->n; i++)
    {
        if (!(l->l_shipdate[i] > cutoff)) continue;
        int64_t k = l->l_orderkey[i];
        uint64_t j = hmix64((uint64_t) k) & t.mask;
        while (t.key[j] != EMPTY_KEY && t.key[j] != k) j = (j + 1) & t.mask;
        if (t.key[j] == EMPTY_KEY) continue;
        int64_t price_c = (int64_t) llround(l->l_extendedprice[i] * 100.0);
        int64_t disc_c = (int64_t) llround(l->l_discount[i] * 100.0);
        int64_t add = price_c * (100 - disc_c);
#ifdef _OPENMP
        __atomic_fetch_add(&num[j], add, __ATOMIC_RELAXED);
        __atomic_fetch_add((uint64_t *) &t.cnt[j], 1, __ATOMIC_RELAXED);
#else
        num[j] += add;
        t.cnt[j]++;
#endif
    }

    int64_t ng = 0;
    for (uint64_t j = 0; j <= t.mask; j++)
        if (t.key[j] != EMPTY_KEY && t.cnt[j] > 0) ng++;
    orc_q3n_group *g = malloc(sizeof(orc_q3n_group) * (ng ? ng : 1));
    int64_t w = 0;
    for (uint64_t j = 0; j <= t.mask; j++)
        if (t.key[j] != EMPTY_KEY && t.cnt[j] > 0)
        {
            g[w].l_orderkey = t.key[j];
            g[w].o_orderdate = t.odate[j];
            g[w].o_shippriority = t.oprio[j];
            g[w].revenue_num = num[j];
            g[w].nitems = t.cnt[j];
            w++;
        }
    qsort(g, ng, sizeof(orc_q3n_group), q3n_group_cmp);
    free(cust.keys);
    free(t.key); free(t.odate); free(t.oprio); free(t.rev); free(t.cnt);
    free(num);
    *out = g;
    return ng;
}

static int q3_group_cmp(const void *a, const void *b)
{
    const orc_q3_group *x = a, *y = b;
    return (x->l_orderkey > y->l_orderkey) - (x->l_orderkey < y->l_orderkey);
}

int64_t orc_q3(const orc_customer *c, const orc_orders *o,
               const orc_lineitem *l, int32_t cutoff, orc_q3_group **out)
{
    /* 1. customer: filter c_mktsegment='BUILDING' → key set
     * (phases are OpenMP-parallel when built with -fopenmp; inserts use CAS,
     * f64 aggregation order varies with thread count — within the judged
     * 1e-6 tolerance, like the GPU path) */
    set64 cust;
    set_init(&cust, c->n + 16);   /* worst case: every customer qualifies */
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (int64_t i = 0; i < c->n; i++)
        if (c->c_mktsegment[i] == 0)
            set_insert(&cust, c->c_custkey[i]);

    /* 2. orders: filter o_orderdate < cutoff, semijoin customer, build table */
    q3tab t;
    tab_init(&t, o->n + 16);      /* worst case: every order qualifies */
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (int64_t i = 0; i < o->n; i++)
    {
        if (!(o->o_orderdate[i] < cutoff)) continue;
        if (!set_contains(&cust, o->o_custkey[i])) continue;
        int64_t k = o->o_orderkey[i];
        uint64_t j = hmix64((uint64_t) k) & t.mask;
        while (1)
        {
            int64_t expect = EMPTY_KEY;
            int64_t cur = __atomic_load_n(&t.key[j], __ATOMIC_RELAXED);
            if (cur == k) break;  /* unique keys: no-op */
            if (cur == EMPTY_KEY &&
                __atomic_compare_exchange_n(&t.key[j], &expect, k, 0,
                                            __ATOMIC_ACQ_REL, __ATOMIC_ACQUIRE))
            {
                t.odate[j] = o->o_orderdate[i];
                t.oprio[j] = o->o_shippriority[i];
                break;
            }
            if (__atomic_load_n(&t.key[j], __ATOMIC_RELAXED) == k) break;
            j = (j + 1) & t.mask;
        }
    }

    /* 3. lineitem: filter l_shipdate > cutoff, probe, aggregate */
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (int64_t i = 0; i < l->n; i++)
    {
        if (!(l->l_shipdate[i] > cutoff)) continue;
        int64_t k = l->l_orderkey[i];
        uint64_t j = hmix64((uint64_t) k) & t.mask;
        while (t.key[j] != EMPTY_KEY && t.key[j] != k) j = (j + 1) & t.mask;
        if (t.key[j] == EMPTY_KEY) continue;
#ifdef _OPENMP
        /* f64 atomic add via CAS loop */
        double add = l->l_extendedprice[i] * (1.0 - l->l_discount[i]);
        uint64_t expect = __atomic_load_n((uint64_t *) &t.rev[j], __ATOMIC_RELAXED);
        while (1)
        {
            double cur, nv;
            uint64_t desired;
            memcpy(&cur, &expect, 8);
            nv = cur + add;
            memcpy(&desired, &nv, 8);
            if (__atomic_compare_exchange_n((uint64_t *) &t.rev[j], &expect,
                                            desired, 0, __ATOMIC_ACQ_REL,
                                            __ATOMIC_ACQUIRE))
                break;
        }
        __atomic_fetch_add((uint64_t *) &t.cnt[j], 1, __ATOMIC_RELAXED);
#else
        t.rev[j] += l->l_extendedprice[i] * (1.0 - l->l_discount[i]);
        t.cnt[j]++;
#endif
    }

    /* 4. extract groups with ≥1 joined lineitem, sorted by l_orderkey */
    int64_t ng = 0;
    for (uint64_t j = 0; j <= t.mask; j++)
        if (t.key[j] != EMPTY_KEY && t.cnt[j] > 0) ng++;
    orc_q3_group *g = malloc(sizeof(orc_q3_group) * (ng ? ng : 1));
    int64_t w = 0;
    for (uint64_t j = 0; j <= t.mask; j++)
        if (t.key[j] != EMPTY_KEY && t.cnt[j] > 0)
        {
            g[w].l_orderkey = t.key[j];
            g[w].o_orderdate = t.odate[j];
            g[w].o_shippriority = t.oprio[j];
            g[w].revenue = t.rev[j];
            g[w].nitems = t.cnt[j];
            w++;
        }
    qsort(g, ng, sizeof(orc_q3_group), q3_group_cmp);

    free(cust.keys);
    free(t.key); free(t.odate); free(t.oprio); free(t.rev); free(t.cnt);
    *out = g;
    return ng;
}


/* ======================================================================
 * Varlena (text-like) Orig codec — restatement of the writer's
 * variable-length path (datumstreamblock.c:1620-1720) and the reader's
 * VARSIZE_ANY walk (datumstreamblock.h:1509-1545):
 *   payload <= 126 B  -> 1-byte short header ((len+1)<<1 | 1), UNALIGNED
 *   larger            -> zero-pad to 4-byte alignment (att_align_zero),
 *                        4-byte header ((len+4)<<2) + payload
 *   the zero padding of a value that then FAILS the capacity check stays
 *   in the emitted block (the writer pads datump before OrigHasSpace)
 * Nulls as in fixed-width Orig.  Byte-exact vs the compiled reference
 * writer (refw_encode_varlena) in tests/test_oracle_cpu.py.
 * ====================================================================== */

/* Dense_Enhanced varlena content walk: null bits per non-repeat slot,
 * compress bit per physical item, varlena datums with zero-pad alignment.
 * Writes into the caller's running payload cursor *w_io. */
static int64_t decode_dense_varlena_content(const uint8_t *c, int64_t logical,
                                            uint8_t *out_payload,
                                            int64_t payload_cap,
                                            int64_t *out_offsets,
                                            uint8_t *out_validity,
                                            int64_t row0, int64_t *w_io)
{
    int16_t flags;
    int32_t hlogical, phys, psize;
    memcpy(&flags, c + 2, 2);
    memcpy(&hlogical, c + 4, 4);
    memcpy(&phys, c + 8, 4);
    memcpy(&psize, c + 12, 4);
    if (hlogical != logical) return -1;
    int has_null = (flags & 0x1) != 0;
    int rle = (flags & 0x2) != 0;
    if (flags & 0x4) return -1;              /* no delta for varlena */
    if (has_null && out_validity == NULL) return -1;
    const uint8_t *p = c + 16;
    int32_t bmbits = 0, csize = 0;
    int32_t nullbits = has_null ? (int32_t) logical : 0;
    if (rle)
    {
        int32_t norepeats;
        memcpy(&norepeats, p, 4);
        memcpy(&bmbits, p + 4, 4);
        memcpy(&csize, p + 12, 4);
        if (has_null) nullbits = norepeats;
        else if (norepeats != 0) return -1;
        p += 16;
    }
    const uint8_t *nbmp = NULL, *bmp = NULL, *cnts = NULL;
    if (has_null) { nbmp = p; p += (nullbits + 7) >> 3; }
    if (rle) { bmp = p; p += (bmbits + 7) >> 3; cnts = p; p += csize; }
    int32_t hdr = (int32_t) (p - c);
    const uint8_t *d0 = c + ((hdr + 7) & ~7);
    const uint8_t *dp = d0, *dend = d0 + psize;

    int64_t lw = *w_io, out = 0;
    int32_t item = 0, coff = 0, npos = 0;
    while (out < logical)
    {
        if (has_null)
        {
            if (npos >= nullbits) return -1;
            int nb = (nbmp[npos >> 3] >> (npos & 7)) & 1;
            npos++;
            if (nb)
            {
                out_validity[row0 + out] = 0;
                out_offsets[row0 + out + 1] = lw;
                out++;
                continue;
            }
        }
        if (rle && item >= bmbits) return -1;
        if (dp < dend && *dp == 0)
            dp = d0 + (((dp - d0) + 3) & ~(int64_t) 3);
        if (dp >= dend) return -1;
        int64_t len;
        const uint8_t *data;
        if (*dp & 1)
        {
            len = (int64_t) (*dp >> 1) - 1;
            data = dp + 1;
            dp += 1 + len;
        }
        else
        {
            uint32_t hdr4;
            if (dp + 4 > dend) return -1;
            memcpy(&hdr4, dp, 4);
            len = (int64_t) (hdr4 >> 2) - 4;
            data = dp + 4;
            dp += 4 + len;
        }
        if (len < 0 || dp > dend) return -1;
        int64_t reps = 1;
        if (rle && (bmp[item >> 3] & (1u << (item & 7))))
        {
            int32_t vlen, v = varint_decode(cnts + coff, &vlen);
            coff += vlen;
            reps += v;
        }
        if (out + reps > logical) return -1;
        if (lw + reps * len > payload_cap) return -3;
        for (int64_t r = 0; r < reps; r++)
        {
            memcpy(out_payload + lw, data, (size_t) len);
            lw += len;
            if (out_validity) out_validity[row0 + out] = 1;
            out_offsets[row0 + out + 1] = lw;
            out++;
        }
        item++;
    }
    if (rle && (coff != csize || item != bmbits)) return -1;
    if (has_null && npos != nullbits) return -1;
    *w_io = lw;
    return out;
}

int64_t orc_aocs_encode_varlena(const uint8_t *payload, const int64_t *offsets,
                                const uint8_t *nulls, int64_t nrows,
                                int64_t first_rownum, int32_t blocksize,
                                uint8_t *out, int64_t outcap)
{
    int32_t maxdata = blocksize - 24;
    int64_t off = 0, row = 0;
    uint8_t *dvals = malloc((size_t) maxdata + 16);
    uint8_t *nbm = malloc((size_t) ((16383 + 7) >> 3) + 8);
    if (!dvals || !nbm) { free(dvals); free(nbm); return -1; }

    while (row < nrows)
    {
        int32_t nth = 0, always = 0;
        int has_null = 0;
        int64_t used = 0;
        memset(nbm, 0, (size_t) ((16383 + 7) >> 3) + 8);
        while (row + nth < nrows && nth + 1 < 16383)
        {
            int isnull = nulls != NULL && nulls[row + nth] != 0;
            int32_t nullsize =
                (isnull || has_null)
                    ? (int32_t) ((((always + 1 + 7) >> 3) + 7) & ~7) : 0;
            if (isnull)
            {
                if (!(16 + nullsize + used + 0 < maxdata))
                    break;
                has_null = 1;
                nbm[always >> 3] |= (uint8_t) (1u << (always & 7));
            }
            else
            {
                int64_t len = offsets[row + nth + 1] - offsets[row + nth];
                int64_t sz, pad = 0;
                if (len + 1 <= 0x7F)
                    sz = len + 1;                 /* short form */
                else
                {
                    pad = (-used) & 3;            /* att_align_zero to 4 */
                    sz = len + 4;
                }
                if (!(16 + nullsize + (used + pad) + sz < maxdata))
                {
                    /* the writer pads BEFORE the failed check; the pad
                     * stays in this block (only the 4B-header case pads) */
                    if (pad && used > 0)
                    {
                        memset(dvals + used, 0, (size_t) pad);
                        used += pad;
                    }
                    break;
                }
                if (pad)
                {
                    memset(dvals + used, 0, (size_t) pad);
                    used += pad;
                }
                if (len + 1 <= 0x7F)
                {
                    dvals[used] = (uint8_t) (((len + 1) << 1) | 1);
                    memcpy(dvals + used + 1, payload + offsets[row + nth], len);
                    used += sz;
                }
                else
                {
                    uint32_t hdr = (uint32_t) ((len + 4) << 2);
                    memcpy(dvals + used, &hdr, 4);
                    memcpy(dvals + used + 4, payload + offsets[row + nth], len);
                    used += sz;
                }
            }
            always++;
            nth++;
        }
        if (nth == 0) { free(dvals); free(nbm); return -1; }

        int32_t nullsz = has_null
            ? (int32_t) ((((nth + 7) >> 3) + 7) & ~7) : 0;
        int32_t content = 16 + nullsz + (int32_t) used;
        int64_t blocklen = (24 + content + 7) & ~7LL;
        if (off + blocklen > outcap) { free(dvals); free(nbm); return -1; }
        uint8_t *blk = out + off;
        memset(blk, 0, (size_t) blocklen);
        uint32_t b03 = (1u << 28) | (1u << 27) | (1u << 24) |
                       (0x00FFFC00u & ((uint32_t) nth << 10)) |
                       (((uint32_t) content >> 11) & 0x3FFu);
        uint32_t b47 = (((uint32_t) content & 0x7FFu) << 21);
        put_u32le(blk, b03);
        put_u32le(blk + 4, b47);
        int64_t frn = first_rownum + row;
        memcpy(blk + 16, &frn, 8);
        uint8_t *c = blk + 24;
        int16_t v16 = 0;  memcpy(c, &v16, 2);
        v16 = has_null ? 1 : 0; memcpy(c + 2, &v16, 2);
        v16 = (int16_t) nth; memcpy(c + 4, &v16, 2);
        v16 = 0; memcpy(c + 6, &v16, 2);
        int32_t v32 = nullsz; memcpy(c + 8, &v32, 4);
        v32 = (int32_t) used; memcpy(c + 12, &v32, 4);
        if (has_null)
            memcpy(c + 16, nbm, (size_t) ((nth + 7) >> 3));
        memcpy(c + 16 + nullsz, dvals, (size_t) used);
        put_u32le(blk + 8, orc_crc32c(0xFFFFFFFFu, blk + 16, blocklen - 16));
        put_u32le(blk + 12, orc_crc32c(0xFFFFFFFFu, blk, 12));
        off += blocklen;
        row += nth;
    }
    free(dvals); free(nbm);
    return off;
}

/* decode: fills out_payload/out_offsets (exclusive, [nrows+1]) and
 * optionally out_validity; null rows have length 0.  Returns rows or
 * -1 / -2 (checksum). */
int64_t orc_aocs_decode_varlena(const uint8_t *stream, int64_t nbytes,
                                int64_t nrows,
                                uint8_t *out_payload, int64_t payload_cap,
                                int64_t *out_offsets, uint8_t *out_validity,
                                int verify_checksums)
{
    int64_t off = 0, row = 0, w = 0;
    out_offsets[0] = 0;
    while (off + 24 <= nbytes)
    {
        uint32_t b03, b47;
        memcpy(&b03, stream + off, 4);
        memcpy(&b47, stream + off + 4, 4);
        if (b03 == 0 && b47 == 0) break;
        uint32_t kind = (b03 >> 28) & 7;
        uint32_t rows, datalen;
        if (kind == 1)
        {
            rows = (b03 & 0x00FFFC00u) >> 10;
            datalen = ((b03 & 0x3FFu) << 11) | ((b47 & 0xFFE00000u) >> 21);
            if ((b47 & 0x1FFFFFu) != 0) return -1; /* no bulk compression */
        }
        else if (kind == 3)                        /* NonBulkDense (RLE) */
        {
            rows = b47 & 0x3FFFFFFFu;
            datalen = b03 & 0x1FFFFFu;
        }
        else
            return -1;
        if (!((b03 >> 27) & 1)) return -1;
        int64_t blocklen = (24 + (int64_t) datalen + 7) & ~7LL;
        if (off + blocklen > nbytes) return -1;
        if (verify_checksums)
        {
            uint32_t bc, hc;
            memcpy(&bc, stream + off + 8, 4);
            memcpy(&hc, stream + off + 12, 4);
            if (hc != orc_crc32c(0xFFFFFFFFu, stream + off, 12)) return -2;
            if (bc != orc_crc32c(0xFFFFFFFFu, stream + off + 16, blocklen - 16)) return -2;
        }
        const uint8_t *c = stream + off + 24;
        int16_t version, flags, nd;
        int32_t nullsz, sz;
        memcpy(&version, c, 2);
        memcpy(&flags, c + 2, 2);
        memcpy(&nd, c + 4, 2);
        memcpy(&nullsz, c + 8, 4);
        memcpy(&sz, c + 12, 4);
        if (version == 1 || version == 2)
        {
            int64_t got = decode_dense_varlena_content(
                c, (int64_t) rows, out_payload, payload_cap,
                out_offsets, out_validity, row, &w);
            if (got == -3) return -3;
            if (got < 0 || got != (int64_t) rows) return -1;
            row += got;
            off += blocklen;
            continue;
        }
        if (version != 0) return -1;
        if ((flags & 1) && out_validity == NULL) return -1;
        if (row + nd > nrows) return -1;
        const uint8_t *nbmp = c + 16;
        const uint8_t *p = c + 16 + nullsz;
        const uint8_t *pend = p + sz;
        for (int32_t r = 0; r < nd; r++)
        {
            if ((flags & 1) && ((nbmp[r >> 3] >> (r & 7)) & 1))
            {
                if (out_validity) out_validity[row + r] = 0;
                out_offsets[row + r + 1] = w;
                continue;
            }
            if (out_validity) out_validity[row + r] = 1;
            /* skip zero padding (reader: *p == 0 -> align_nominal) */
            if (p < pend && *p == 0)
                p = c + 16 + nullsz +
                    (((p - (c + 16 + nullsz)) + 3) & ~(int64_t) 3);
            if (p >= pend) return -1;
            int64_t len;
            const uint8_t *data;
            if (*p & 1)
            {
                len = (int64_t) (*p >> 1) - 1;
                data = p + 1;
                p += 1 + len;
            }
            else
            {
                uint32_t hdr;
                if (p + 4 > pend) return -1;
                memcpy(&hdr, p, 4);
                len = (int64_t) (hdr >> 2) - 4;
                data = p + 4;
                p += 4 + len;
            }
            if (len < 0 || p > pend) return -1;
            if (w + len > payload_cap) return -3;
            memcpy(out_payload + w, data, (size_t) len);
            w += len;
            out_offsets[row + r + 1] = w;
        }
        row += nd;
        off += blocklen;
    }
    return row;
}
