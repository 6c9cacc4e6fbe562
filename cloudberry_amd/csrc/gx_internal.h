/*
 * gx_internal.h — shared host/device inline contract for libgpuexec.so.
 *
 * The functions here are the GPU-side counterparts of the oracle's
 * restatements (oracle/oracle.c) and are pinned by the same golden vectors
 * through the parity tests.  Reference citations per function.
 */
#ifndef GX_INTERNAL_H
#define GX_INTERNAL_H

#include <stdint.h>
#include <hip/hip_runtime.h>

#define GX_HD __host__ __device__ __forceinline__

/* ---------- bit-exact reference hashes ---------- */

GX_HD uint32_t gx_rot32(uint32_t x, int k) { return (x << k) | (x >> (32 - k)); }

/* Jenkins final() — src/common/hashfn.c:133-142 */
GX_HD uint32_t gx_hash_bytes_uint32(uint32_t k)
{
    uint32_t a, b, c;
    a = b = c = 0x9e3779b9u + 4u + 3923095u;
    a += k;
    c ^= b; c -= gx_rot32(b, 14);
    a ^= c; a -= gx_rot32(c, 11);
    b ^= a; b -= gx_rot32(a, 25);
    c ^= b; c -= gx_rot32(b, 16);
    a ^= c; a -= gx_rot32(c, 4);
    b ^= a; b -= gx_rot32(a, 14);
    c ^= b; c -= gx_rot32(b, 24);
    return c;
}

/* hashint8 — src/backend/access/hash/hashfunc.c:85-101 */
GX_HD uint32_t gx_hashint8(int64_t val)
{
    uint32_t lohalf = (uint32_t) val;
    uint32_t hihalf = (uint32_t) ((uint64_t) val >> 32);
    lohalf ^= (val >= 0) ? hihalf : ~hihalf;
    return gx_hash_bytes_uint32(lohalf);
}

/* single-key cdbhash chain — cdb/cdbhash.c:171-247 (init 0, rot1, xor) */
GX_HD uint32_t gx_cdbhash_i64(int64_t v) { return gx_hashint8(v); }

/* jump_consistent_hash — cdb/cdbhash.c:530-541.  f64 ops are IEEE on both
 * host and gfx950 → bit-identical routing. */
GX_HD int32_t gx_jump_consistent_hash(uint64_t key, int32_t num_segments)
{
    int64_t b = -1, j = 0;
    while (j < num_segments)
    {
        b = j;
        key = key * 2862933555777941757ULL + 1;
        j = (int64_t) ((double) (b + 1) *
                       ((double) (1LL << 31) / (double) ((key >> 33) + 1)));
    }
    return (int32_t) b;
}

/* Motion routing — nodeMotion.c:1088 + cdbhash.c:253-285 (REDUCE_JUMP_HASH) */
GX_HD int32_t gx_route_i64(int64_t key, int32_t nsegs)
{
    return gx_jump_consistent_hash((uint64_t) gx_cdbhash_i64(key), nsegs);
}

/* hashint4 — hashfunc.c:73-77 */
GX_HD uint32_t gx_hashint4(int32_t val)
{
    return gx_hash_bytes_uint32((uint32_t) val);
}

/* N-attribute cdbhash chain — cdbhash.c:171-247: per attribute
 * rotate-left-1 then XOR the type hash; NULL contributes the rotation only.
 * types[k]: 0 = int8 (hashint8), 1 = int4/int2/date (hashint4). */
GX_HD uint32_t gx_cdbhash_multi(const int64_t *vals, const uint8_t *isnull,
                                const int32_t *types, int32_t nkeys)
{
    uint32_t hashkey = 0;                                  /* cdbhashinit */
    for (int32_t k = 0; k < nkeys; k++)
    {
        hashkey = (hashkey << 1) | (hashkey >> 31);
        if (isnull && isnull[k]) continue;
        hashkey ^= (types && types[k] == 1)
                       ? gx_hashint4((int32_t) vals[k])
                       : gx_hashint8(vals[k]);
    }
    return hashkey;
}

/* ---------- internal hash-table hash (NOT parity-relevant) ---------- */
GX_HD uint64_t gx_hmix64(uint64_t x)
{
    x ^= x >> 33; x *= 0xFF51AFD7ED558CCDULL;
    x ^= x >> 33; x *= 0xC4CEB9FE1A85EC53ULL;
    x ^= x >> 33; return x;
}

/* ---------- deterministic synthetic data (contract == oracle/oracle.c) ---------- */

GX_HD uint64_t gx_splitmix64(uint64_t x)
{
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

GX_HD uint64_t gx_mix(uint64_t seed, uint64_t stream, uint64_t idx)
{
    return gx_splitmix64(gx_splitmix64(seed ^ (stream * 0xA24BAED4963EE407ULL)) + idx);
}

enum {
    GX_ST_CUST_SEG = 1, GX_ST_ORD_CUST = 2, GX_ST_ORD_DATE = 3,
    GX_ST_ORD_PRIO = 4, GX_ST_LI_COUNT = 5, GX_ST_LI_SHIP = 6,
    GX_ST_LI_PRICE = 7, GX_ST_LI_DISC = 8, GX_ST_LI_FLAG = 9,
    GX_ST_LI_STATUS = 10,
};

/* DateADT constants (validated against oracle orc_date_adt in tests):
 * 1992-01-01 = -2922, 1992-01-02 = -2921 (days since 2000-01-01) */
#define GX_ORD_DATE_LO (-2922)
#define GX_ORD_DATE_SPAN 2405
#define GX_LI_DATE_LO (-2921)
#define GX_LI_DATE_SPAN 2525

GX_HD uint8_t gx_gen_mktsegment(uint64_t seed, int64_t i)
{ return (uint8_t) (gx_mix(seed, GX_ST_CUST_SEG, (uint64_t) i) % 5); }

GX_HD int64_t gx_gen_ocustkey(uint64_t seed, int64_t i, int64_t ncust)
{
    int64_t pool = (ncust * 2) / 3;
    if (pool < 1) pool = 1;
    return 1 + (int64_t) (gx_mix(seed, GX_ST_ORD_CUST, (uint64_t) i) % (uint64_t) pool);
}
GX_HD int32_t gx_gen_odate(uint64_t seed, int64_t i)
{ return GX_ORD_DATE_LO + (int32_t) (gx_mix(seed, GX_ST_ORD_DATE, (uint64_t) i) % (GX_ORD_DATE_SPAN + 1)); }
GX_HD int32_t gx_gen_oprio(uint64_t seed, int64_t i)
{ return (int32_t) (gx_mix(seed, GX_ST_ORD_PRIO, (uint64_t) i) % 5); }
GX_HD int32_t gx_gen_nlines(uint64_t seed, int64_t okey)
{ return 1 + (int32_t) (gx_mix(seed, GX_ST_LI_COUNT, (uint64_t) okey) % 7); }
GX_HD int32_t gx_gen_shipdate(uint64_t seed, int64_t okey, int32_t line)
{ return GX_LI_DATE_LO + (int32_t) (gx_mix(seed, GX_ST_LI_SHIP, (uint64_t) okey * 8 + line) % (GX_LI_DATE_SPAN + 1)); }
GX_HD double gx_gen_price(uint64_t seed, int64_t okey, int32_t line)
{ return (double) (90000 + gx_mix(seed, GX_ST_LI_PRICE, (uint64_t) okey * 8 + line) % 10410001ULL) / 100.0; }
GX_HD uint8_t gx_gen_returnflag(uint64_t seed, int64_t okey, int32_t line)
{ return (uint8_t) (gx_mix(seed, GX_ST_LI_FLAG, (uint64_t) okey * 8 + line) % 3); }
GX_HD uint8_t gx_gen_linestatus(uint64_t seed, int64_t okey, int32_t line)
{ return (uint8_t) (gx_mix(seed, GX_ST_LI_STATUS, (uint64_t) okey * 8 + line) % 2); }
GX_HD double gx_gen_discount(uint64_t seed, int64_t okey, int32_t line)
{ return (double) (gx_mix(seed, GX_ST_LI_DISC, (uint64_t) okey * 8 + line) % 11) / 100.0; }

/* ---------- AOCS geometry (format: DESIGN.md "Data layout in HBM") ---------- */

/* reference writer capacity rule, datumstreamblock.c:1508-1560 */
GX_HD int32_t gx_aocs_rows_per_block(int width, int32_t blocksize)
{
    int32_t maxdata = blocksize - 24;
    int32_t n = 0;
    while ((n + 1 < 16383) && (16 + n * width + width < maxdata))
        n++;
    return n;
}

/* whole-block length for a given row count (RoundUp8, cdbappendonlystorage.h:39) */
GX_HD int64_t gx_aocs_block_len(int width, int64_t rows)
{
    return (24 + 16 + rows * width + 7) & ~(int64_t) 7;
}

/* datum offset inside a block: 8 hdr + 8 crc + 8 firstRowNum + 16 Orig hdr */
#define GX_AOCS_DATUM_OFF 40

/* O(1) stream addressing for full-block columns (all blocks before the last
 * hold exactly rpb rows — guaranteed by our writer) */
struct gx_colmeta {
    int32_t width;
    int32_t rpb;               /* rows per full block */
    int64_t nrows;
    int64_t full_block_len;    /* gx_aocs_block_len(width, rpb) */
    int64_t nbytes;            /* whole stream */
    uint64_t magic;            /* floor(2^64/rpb)+1 — division-free row→block */
};

/* scan-time tuple visibility (aocs_getnext → AppendOnlyVisimap_IsVisible) */
GX_HD bool gx_vm_hidden(const uint8_t *vm, int64_t row)
{
    return vm != nullptr && ((vm[row >> 3] >> (row & 7)) & 1);
}

/* fill the magic multiplier: q = mulhi64(row, magic) == row / rpb, exact for
 * row < 2^64/rpb (rpb ≤ 16382 → exact beyond 10^15 rows; verified in tests
 * against integer division over boundary rows) */
GX_HD void gx_colmeta_finish(gx_colmeta *m)
{
    m->magic = (~0ULL) / (uint64_t) m->rpb + 1;
}

GX_HD uint64_t gx_mulhi64(uint64_t a, uint64_t b)
{
#ifdef __HIP_DEVICE_COMPILE__
    return __umul64hi(a, b);
#else
    return (uint64_t) (((unsigned __int128) a * b) >> 64);
#endif
}

/* non-temporal variant: stream data is read once — bypassing cache keeps
 * L2/L3 for the join table (A/B experiment; device-only) */
template <typename T>
__device__ __forceinline__ T gx_col_get_nt(const uint8_t *stream,
                                           const gx_colmeta m, int64_t row)
{
    int64_t b = (int64_t) gx_mulhi64((uint64_t) row, m.magic);
    int64_t r = row - b * m.rpb;
    return __builtin_nontemporal_load(
        (const T *) (stream + b * m.full_block_len + GX_AOCS_DATUM_OFF +
                     r * (int64_t) sizeof(T)));
}

template <typename T>
GX_HD T gx_col_get(const uint8_t *stream, const gx_colmeta m, int64_t row)
{
    int64_t b = (int64_t) gx_mulhi64((uint64_t) row, m.magic);
    int64_t r = row - b * m.rpb;
    return *(const T *) (stream + b * m.full_block_len + GX_AOCS_DATUM_OFF +
                         r * (int64_t) sizeof(T));
}

/* filter compare (gx_filter.op): 0 '<',1 '>',2 '=',3 '!=',4 '<=',5 '>=' */
template <typename T>
GX_HD bool gx_cmp(int op, T v, T lit)
{
    switch (op)
    {
        case 0: return v < lit;
        case 1: return v > lit;
        case 2: return v == lit;
        case 3: return v != lit;
        case 4: return v <= lit;
        default: return v >= lit;
    }
}

/* per-block directory entry for variable-geometry (RLE/Dense) streams */
struct gx_blockref {
    int64_t offset;              /* byte offset of the AO block */
    int64_t first_row;           /* 0-based logical first row */
    int32_t rows;                /* logical rows in the block */
    int32_t pad;
};

/* gx_ord_row (Motion-1 payload) is declared in include/gpuexec.h — it is
 * also a test-ABI type. */

struct gx_qual_row {             /* Motion-2 payload: qualifying orders */
    int64_t okey;
    int32_t odate;
    int32_t oprio;
};

#endif /* GX_INTERNAL_H */
