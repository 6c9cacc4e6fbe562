/*
 * ref_writer.c — drive the REFERENCE's own datum-stream block writer
 * (src/backend/utils/datumstream/datumstreamblock.c, compiled standalone
 * where it lies — see oracle/Makefile) to produce REAL
 * Dense_Enhanced RLE/DELTA block content for decoder parity tests.
 *
 * TEST INFRASTRUCTURE ONLY (oracle/ usage contract, oracle/oracle.h).
 * The stubs below replace the server runtime the writer expects: palloc →
 * malloc, elog → print-and-abort on ERROR, GUC debug flags off.  No
 * reference sources are copied; this file only CALLS the compiled object.
 */
#include "postgres.h"
#include "storage/relfilelocator.h"
#include "storage/relfilenode.h"
#include "utils/datumstreamblock.h"

/* from utils/datumstream.h (not includable standalone — deep AM deps) */
#define MAXDATUM_PER_AOCS_ORIG_BLOCK AOSmallContentHeader_MaxRowCount
#define INITIALDATUM_PER_AOCS_DENSE_BLOCK AOSmallContentHeader_MaxRowCount
#define MAXDATUM_PER_AOCS_DENSE_BLOCK AONonBulkDenseContentHeader_MaxLargeRowCount
#include "cdb/cdbappendonlystorage.h"

#include <stdio.h>
#include <stdarg.h>
#include <stdlib.h>
#include <string.h>

/* ---------------- server-runtime stubs ---------------- */

int FileEncryptionEnabled = 0;   /* extern int — access/xlog.h:84 */
bool Debug_appendonly_print_insert = false;
bool Debug_appendonly_print_insert_tuple = false;
bool Debug_appendonly_print_scan = false;
bool Debug_datumstream_block_read_check_integrity = false;
bool Debug_datumstream_block_write_check_integrity = false;
bool Debug_datumstream_write_print_small_varlena_info = false;
bool Debug_datumstream_write_use_small_initial_buffers = false;

MemoryContext CurrentMemoryContext = NULL;

MemoryContext
MemoryContextSwitchTo(MemoryContext context)
{
    (void) context;
    return NULL;
}

void *
palloc(Size size)
{
    return malloc(size);
}

void
pfree(void *pointer)
{
    free(pointer);
}

bool
errstart(int elevel, const char *domain)
{
    (void) domain;
    return elevel >= ERROR;
}

void
errfinish(const char *filename, int lineno, const char *funcname)
{
    fprintf(stderr, "ref_writer: ereport(ERROR) at %s:%d in %s\n",
            filename ? filename : "?", lineno, funcname ? funcname : "?");
    abort();
}

void
errmsg(const char *fmt,...)
{
    (void) fmt;
}

void
errmsg_internal(const char *fmt,...)
{
    (void) fmt;
}

void
errdetail_internal(const char *fmt,...)
{
    (void) fmt;
}

#undef vsprintf
#undef vfprintf
#undef fprintf
#undef printf

int
pg_vsprintf(char *str, const char *fmt, va_list args)
{
    return vsprintf(str, fmt, args);
}

int
pg_fprintf(FILE *stream, const char *fmt,...)
{
    va_list ap;
    int n;

    va_start(ap, fmt);
    n = vfprintf(stream, fmt, ap);
    va_end(ap);
    return n;
}

int
pg_sprintf(char *str, const char *fmt,...)
{
    va_list ap;
    int n;

    va_start(ap, fmt);
    n = vsprintf(str, fmt, ap);
    va_end(ap);
    return n;
}

void
varattrib_untoast_ptr_len(Datum d, char **datastart, int *len, void **tofree)
{
    /* restatement of detoast.c:652-705 for PLAIN (non-TOASTed) datums —
     * the only kind this harness feeds */
    struct varlena *va = (struct varlena *) DatumGetPointer(d);

    *tofree = NULL;
    if (va == NULL)
        abort();
    if (VARATT_IS_SHORT(va))
    {
        *len = VARSIZE_SHORT(va) - VARHDRSZ_SHORT;
        *datastart = VARDATA_SHORT(va);
        return;
    }
    if (VARATT_IS_EXTENDED(va))
        abort();                /* TOAST pointers never reach this harness */
    *datastart = VARDATA(va);
    *len = VARSIZE(va) - VARHDRSZ;
}

void
DecryptAOBlock(unsigned char *dataBuffer, int dataLen, RelFileNode *file_node)
{
    (void) dataBuffer; (void) dataLen; (void) file_node;
    fprintf(stderr, "STUB DecryptAOBlock called\n");
    abort();
}

void
EncryptAOBLock(unsigned char *dataBuffer, int dataLen, RelFileNode *file_node)
{
    (void) dataBuffer; (void) dataLen; (void) file_node;
    fprintf(stderr, "STUB EncryptAOBLock called (FileEncryptionEnabled=%d)\n",
            FileEncryptionEnabled);
    abort();
}

/* ---------------- harness ---------------- */

static int
cb_zero(void *arg)
{
    (void) arg;
    return 0;
}

/*
 * Feed nrows fixed-width values through the reference writer and emit the
 * CONTENT of each datum-stream block back-to-back into out.  Returns the
 * number of blocks, or -1.  block_lens/block_rows must hold max_blocks.
 * version: 0 = Original, 2 = Dense_Enhanced.  rle/delta toggle compression.
 * nulls: optional per-row null flags (NULL = no nulls).
 */
int
refw_encode_nulls(const void *vals, const uint8 *nulls, int width, int64 nrows,
                  int version, int rle, int delta, int32 maxDataBlockSize,
                  uint8 *out, int64 outcap,
                  int32 *block_lens, int32 *block_rows, int max_blocks)
{
    DatumStreamTypeInfo ti;
    DatumStreamBlockWrite dsw;
    RelFileLocator loc;
    const uint8 *src = (const uint8 *) vals;
    int64 off = 0;
    int nblocks = 0;
    int32 rows_in_block = 0;

    memset(&ti, 0, sizeof(ti));
    ti.datumlen = width;
    ti.typid = (width == 8) ? 20 : (width == 4) ? 23 : 18;  /* int8/int4/char */
    ti.typstorage = 'p';
    ti.align = (width == 8) ? 'd' : (width == 4) ? 'i' : 'c';
    ti.byval = true;

    memset(&loc, 0, sizeof(loc));
    memset(&dsw, 0, sizeof(dsw));
    DatumStreamBlockWrite_Init(&dsw, &ti,
                               version == 0 ? DatumStreamVersion_Original
                                            : DatumStreamVersion_Dense_Enhanced,
                               rle != 0, delta != 0,
                               version == 0 ? MAXDATUM_PER_AOCS_ORIG_BLOCK
                                            : INITIALDATUM_PER_AOCS_DENSE_BLOCK,
                               version == 0 ? MAXDATUM_PER_AOCS_ORIG_BLOCK
                                            : MAXDATUM_PER_AOCS_DENSE_BLOCK,
                               maxDataBlockSize,
                               cb_zero, NULL, cb_zero, NULL, &loc);
    DatumStreamBlockWrite_GetReady(&dsw);

    for (int64 i = 0; i < nrows; i++)
    {
        Datum d = 0;
        bool isnull = (nulls != NULL && nulls[i] != 0);
        void *tofree = NULL;

        if (!isnull)
            memcpy(&d, src + i * width, width);
        if (DatumStreamBlockWrite_Put(&dsw, d, isnull, &tofree) < 0)
        {
            int64 len;

            if (nblocks >= max_blocks || off + maxDataBlockSize > outcap)
                return -1;
            len = DatumStreamBlockWrite_Block(&dsw, out + off, &loc);
            block_lens[nblocks] = (int32) len;
            block_rows[nblocks] = rows_in_block;
            nblocks++;
            off += len;
            rows_in_block = 0;
            DatumStreamBlockWrite_GetReady(&dsw);
            if (DatumStreamBlockWrite_Put(&dsw, d, isnull, &tofree) < 0)
                return -1;
        }
        rows_in_block++;
    }
    if (rows_in_block > 0)
    {
        int64 len;

        if (nblocks >= max_blocks || off + maxDataBlockSize > outcap)
            return -1;
        len = DatumStreamBlockWrite_Block(&dsw, out + off, &loc);
        block_lens[nblocks] = (int32) len;
        block_rows[nblocks] = rows_in_block;
        nblocks++;
        off += len;
    }
    return nblocks;
}

/* back-compat entry point: no nulls */
int
refw_encode(const void *vals, int width, int64 nrows,
            int version, int rle, int delta, int32 maxDataBlockSize,
            uint8 *out, int64 outcap,
            int32 *block_lens, int32 *block_rows, int max_blocks)
{
    return refw_encode_nulls(vals, NULL, width, nrows,
                             version, rle, delta, maxDataBlockSize,
                             out, outcap, block_lens, block_rows, max_blocks);
}

/*
 * Varlena (text-like) columns: feed 4-byte-header varlena datums through
 * the reference writer (typid 25, typstorage 'x', align 'i'); the writer
 * itself converts to short form / aligns (datumstreamblock.c:1620-1720).
 * payload = concatenated value bytes, offsets[nrows+1] exclusive.
 */
int
refw_encode_varlena(const uint8 *payload, const int64 *offsets,
                    const uint8 *nulls, int64 nrows,
                    int version, int rle, int32 maxDataBlockSize,
                    uint8 *out, int64 outcap,
                    int32 *block_lens, int32 *block_rows, int max_blocks)
{
    DatumStreamTypeInfo ti;
    DatumStreamBlockWrite dsw;
    RelFileLocator loc;
    int64 off = 0;
    int nblocks = 0;
    int32 rows_in_block = 0;
    uint8 *scratch = malloc(16 * 1024 * 1024);

    if (!scratch)
        return -1;
    memset(&ti, 0, sizeof(ti));
    ti.datumlen = -1;
    ti.typid = 25;              /* text */
    ti.typstorage = 'x';
    ti.align = 'i';
    ti.byval = false;

    memset(&loc, 0, sizeof(loc));
    memset(&dsw, 0, sizeof(dsw));
    DatumStreamBlockWrite_Init(&dsw, &ti,
                               version == 0 ? DatumStreamVersion_Original
                                            : DatumStreamVersion_Dense_Enhanced,
                               rle != 0, /* delta */ false,
                               version == 0 ? MAXDATUM_PER_AOCS_ORIG_BLOCK
                                            : INITIALDATUM_PER_AOCS_DENSE_BLOCK,
                               version == 0 ? MAXDATUM_PER_AOCS_ORIG_BLOCK
                                            : MAXDATUM_PER_AOCS_DENSE_BLOCK,
                               maxDataBlockSize,
                               cb_zero, NULL, cb_zero, NULL, &loc);
    DatumStreamBlockWrite_GetReady(&dsw);

    for (int64 i = 0; i < nrows; i++)
    {
        Datum d = 0;
        bool isnull = (nulls != NULL && nulls[i] != 0);
        void *tofree = NULL;

        if (!isnull)
        {
            int64 len = offsets[i + 1] - offsets[i];

            if (len + 4 > 16 * 1024 * 1024)
            {
                free(scratch);
                return -1;
            }
            SET_VARSIZE(scratch, len + 4);
            memcpy(scratch + 4, payload + offsets[i], len);
            d = PointerGetDatum(scratch);
        }
        if (DatumStreamBlockWrite_Put(&dsw, d, isnull, &tofree) < 0)
        {
            int64 len;

            if (nblocks >= max_blocks || off + maxDataBlockSize > outcap)
            {
                free(scratch);
                return -1;
            }
            len = DatumStreamBlockWrite_Block(&dsw, out + off, &loc);
            block_lens[nblocks] = (int32) len;
            block_rows[nblocks] = rows_in_block;
            nblocks++;
            off += len;
            rows_in_block = 0;
            DatumStreamBlockWrite_GetReady(&dsw);
            if (DatumStreamBlockWrite_Put(&dsw, d, isnull, &tofree) < 0)
            {
                free(scratch);
                return -1;
            }
        }
        rows_in_block++;
    }
    if (rows_in_block > 0)
    {
        int64 len;

        if (nblocks >= max_blocks || off + maxDataBlockSize > outcap)
        {
            free(scratch);
            return -1;
        }
        len = DatumStreamBlockWrite_Block(&dsw, out + off, &loc);
        block_lens[nblocks] = (int32) len;
        block_rows[nblocks] = rows_in_block;
        nblocks++;
        off += len;
    }
    free(scratch);
    return nblocks;
}
