"""ctypes binding for liboracle.so (TEST INFRASTRUCTURE — see oracle/oracle.h:
only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
import this module; it is never the product path)."""
import ctypes
import os

import numpy as np

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_SO = os.path.join(_ROOT, "oracle", "liboracle.so")


class _Customer(ctypes.Structure):
    _fields_ = [("c_custkey", ctypes.POINTER(ctypes.c_int64)),
                ("c_mktsegment", ctypes.POINTER(ctypes.c_uint8)),
                ("n", ctypes.c_int64)]


class _Orders(ctypes.Structure):
    _fields_ = [("o_orderkey", ctypes.POINTER(ctypes.c_int64)),
                ("o_custkey", ctypes.POINTER(ctypes.c_int64)),
                ("o_orderdate", ctypes.POINTER(ctypes.c_int32)),
                ("o_shippriority", ctypes.POINTER(ctypes.c_int32)),
                ("n", ctypes.c_int64)]


class _Lineitem(ctypes.Structure):
    _fields_ = [("l_orderkey", ctypes.POINTER(ctypes.c_int64)),
                ("l_extendedprice", ctypes.POINTER(ctypes.c_double)),
                ("l_discount", ctypes.POINTER(ctypes.c_double)),
                ("l_shipdate", ctypes.POINTER(ctypes.c_int32)),
                ("n", ctypes.c_int64)]


class _GroupN(ctypes.Structure):
    _fields_ = [("l_orderkey", ctypes.c_int64),
                ("o_orderdate", ctypes.c_int32),
                ("o_shippriority", ctypes.c_int32),
                ("revenue_num", ctypes.c_int64),
                ("nitems", ctypes.c_int64)]


class _Group(ctypes.Structure):
    _fields_ = [("l_orderkey", ctypes.c_int64),
                ("o_orderdate", ctypes.c_int32),
                ("o_shippriority", ctypes.c_int32),
                ("revenue", ctypes.c_double),
                ("nitems", ctypes.c_int64)]


def _load():
    if not os.path.exists(_SO):
        raise RuntimeError(f"{_SO} missing — run `make -C oracle` first")
    lib = ctypes.CDLL(_SO)
    lib.orc_hash_bytes_uint32.restype = ctypes.c_uint32
    lib.orc_hash_bytes_uint32.argtypes = [ctypes.c_uint32]
    lib.orc_hashint8.restype = ctypes.c_uint32
    lib.orc_hashint8.argtypes = [ctypes.c_int64]
    lib.orc_cdbhash_i64.restype = ctypes.c_uint32
    lib.orc_cdbhash_i64.argtypes = [ctypes.c_int64]
    lib.orc_jump_consistent_hash.restype = ctypes.c_int32
    lib.orc_jump_consistent_hash.argtypes = [ctypes.c_uint64, ctypes.c_int32]
    lib.orc_route_i64.restype = ctypes.c_int32
    lib.orc_route_i64.argtypes = [ctypes.c_int64, ctypes.c_int32]
    lib.orc_hashint4.restype = ctypes.c_uint32
    lib.orc_hashint4.argtypes = [ctypes.c_int32]
    lib.orc_cdbhash_multi.restype = ctypes.c_uint32
    lib.orc_cdbhash_multi.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                      ctypes.c_void_p, ctypes.c_int32]
    lib.orc_route_multi_batch.restype = None
    lib.orc_route_multi_batch.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                          ctypes.c_void_p, ctypes.c_int32,
                                          ctypes.c_int64, ctypes.c_int32,
                                          ctypes.c_void_p]
    lib.orc_route_i64_batch.restype = None
    lib.orc_route_i64_batch.argtypes = [ctypes.c_void_p, ctypes.c_int64,
                                        ctypes.c_int32, ctypes.c_void_p]
    lib.orc_crc32c.restype = ctypes.c_uint32
    lib.orc_crc32c.argtypes = [ctypes.c_uint32, ctypes.c_void_p, ctypes.c_size_t]
    lib.orc_date_adt.restype = ctypes.c_int32
    lib.orc_date_adt.argtypes = [ctypes.c_int] * 3
    lib.orc_set_threads.restype = ctypes.c_int
    lib.orc_set_threads.argtypes = [ctypes.c_int]
    lib.orc_splitmix64.restype = ctypes.c_uint64
    lib.orc_splitmix64.argtypes = [ctypes.c_uint64]
    lib.orc_mix.restype = ctypes.c_uint64
    lib.orc_mix.argtypes = [ctypes.c_uint64] * 3
    lib.orc_ncustomer.restype = ctypes.c_int64
    lib.orc_ncustomer.argtypes = [ctypes.c_double]
    lib.orc_norders.restype = ctypes.c_int64
    lib.orc_norders.argtypes = [ctypes.c_double]
    for name, st in (("customer", _Customer), ("orders", _Orders), ("lineitem", _Lineitem)):
        fn = getattr(lib, f"orc_gen_{name}")
        fn.restype = ctypes.c_int
        fn.argtypes = [ctypes.c_double, ctypes.c_uint64, ctypes.c_int, ctypes.c_int,
                       ctypes.POINTER(st)]
    lib.orc_aocs_rows_per_block.restype = ctypes.c_int32
    lib.orc_aocs_rows_per_block.argtypes = [ctypes.c_int, ctypes.c_int32]
    lib.orc_aocs_encoded_size.restype = ctypes.c_int64
    lib.orc_aocs_encoded_size.argtypes = [ctypes.c_int, ctypes.c_int64, ctypes.c_int32]
    lib.orc_aocs_encode_rle.restype = ctypes.c_int64
    lib.orc_aocs_encode_rle.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_int64,
                                        ctypes.c_int64, ctypes.c_int32,
                                        ctypes.c_void_p, ctypes.c_int64]
    lib.orc_aocs_encode_rle_delta.restype = ctypes.c_int64
    lib.orc_aocs_encode_rle_delta.argtypes = lib.orc_aocs_encode_rle.argtypes
    lib.orc_aocs_encode_rle_delta_nulls.restype = ctypes.c_int64
    lib.orc_aocs_encode_rle_delta_nulls.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_int64,
        ctypes.c_int64, ctypes.c_int32, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_int64]
    lib.orc_aocs_encode_orig_nulls.restype = ctypes.c_int64
    lib.orc_aocs_encode_orig_nulls.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_int64,
        ctypes.c_int64, ctypes.c_int32, ctypes.c_void_p, ctypes.c_int64]
    lib.orc_aocs_decode_nullable.restype = ctypes.c_int64
    lib.orc_aocs_decode_nullable.argtypes = [
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_int, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_int, ctypes.c_int]
    lib.orc_aocs_encode_varlena.restype = ctypes.c_int64
    lib.orc_aocs_encode_varlena.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
        ctypes.c_int64, ctypes.c_int32, ctypes.c_void_p, ctypes.c_int64]
    lib.orc_aocs_decode_varlena.restype = ctypes.c_int64
    lib.orc_aocs_decode_varlena.argtypes = [
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_int64, ctypes.c_void_p,
        ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int]
    lib.orc_aocs_encode_varlena_rle.restype = ctypes.c_int64
    lib.orc_aocs_encode_varlena_rle.argtypes = lib.orc_aocs_encode_varlena.argtypes
    lib.orc_aocs_encode_zstd.restype = ctypes.c_int64
    lib.orc_aocs_encode_zstd.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                         ctypes.c_int64, ctypes.c_int64,
                                         ctypes.c_int32, ctypes.c_int,
                                         ctypes.c_void_p, ctypes.c_int64]
    lib.orc_aocs_decode_c.restype = ctypes.c_int64
    lib.orc_aocs_decode_c.argtypes = [ctypes.c_void_p, ctypes.c_int64, ctypes.c_int,
                                      ctypes.c_void_p, ctypes.c_int64,
                                      ctypes.c_int, ctypes.c_int]
    lib.orc_aocs_encode_zlib.restype = ctypes.c_int64
    lib.orc_aocs_encode_zlib.argtypes = [ctypes.c_void_p, ctypes.c_int,
                                         ctypes.c_int64, ctypes.c_int64,
                                         ctypes.c_int32, ctypes.c_int,
                                         ctypes.c_void_p, ctypes.c_int64]
    lib.orc_aocs_encode.restype = ctypes.c_int64
    lib.orc_aocs_encode.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_int64,
                                    ctypes.c_int64, ctypes.c_int32,
                                    ctypes.c_void_p, ctypes.c_int64]
    lib.orc_aocs_decode.restype = ctypes.c_int64
    lib.orc_aocs_decode.argtypes = [ctypes.c_void_p, ctypes.c_int64, ctypes.c_int,
                                    ctypes.c_void_p, ctypes.c_int64, ctypes.c_int]
    lib.orc_q3.restype = ctypes.c_int64
    lib.orc_q3.argtypes = [ctypes.POINTER(_Customer), ctypes.POINTER(_Orders),
                           ctypes.POINTER(_Lineitem), ctypes.c_int32,
                           ctypes.POINTER(ctypes.POINTER(_Group))]
    lib.orc_q3_numeric.restype = ctypes.c_int64
    lib.orc_q3_numeric.argtypes = [ctypes.POINTER(_Customer), ctypes.POINTER(_Orders),
                                   ctypes.POINTER(_Lineitem), ctypes.c_int32,
                                   ctypes.POINTER(ctypes.POINTER(_GroupN))]
    return lib


lib = _load()

CUTOFF_19950315 = lib.orc_date_adt(1995, 3, 15)


class _Q1Group(ctypes.Structure):
    _fields_ = [("returnflag", ctypes.c_int8), ("linestatus", ctypes.c_int8),
                ("count", ctypes.c_int64), ("sum_price", ctypes.c_double),
                ("sum_revenue", ctypes.c_double)]


def q1(sf, cutoff, seed=42):
    lib.orc_q1.restype = ctypes.c_int
    lib.orc_q1.argtypes = [ctypes.c_double, ctypes.c_uint64, ctypes.c_int32,
                           ctypes.POINTER(_Q1Group)]
    out = (_Q1Group * 6)()
    assert lib.orc_q1(sf, seed, cutoff, out) == 0
    return {"count": np.array([g.count for g in out], np.int64),
            "sum_price": np.array([g.sum_price for g in out]),
            "sum_revenue": np.array([g.sum_revenue for g in out])}


_refw = None


def ref_writer():
    """The REFERENCE's own datum-stream writer (oracle/_ref/libpgwriter.so,
    compiled from /root/reference sources) — None when not built."""
    global _refw
    if _refw is None:
        so = os.path.join(_ROOT, "oracle", "_ref", "libpgwriter.so")
        if not os.path.exists(so):
            return None
        _refw = ctypes.CDLL(so)
        _refw.refw_encode.restype = ctypes.c_int
        _refw.refw_encode.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_int64,
                                      ctypes.c_int, ctypes.c_int, ctypes.c_int,
                                      ctypes.c_int32, ctypes.c_void_p, ctypes.c_int64,
                                      ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int]
        _refw.refw_encode_nulls.restype = ctypes.c_int
        _refw.refw_encode_nulls.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_int64,
            ctypes.c_int, ctypes.c_int, ctypes.c_int,
            ctypes.c_int32, ctypes.c_void_p, ctypes.c_int64,
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int]
    return _refw


def ref_writer_stream(vals, version=2, rle=1, delta=1, blocksize=32768,
                      nulls=None):
    """Run the reference writer and wrap its content blocks in our AO
    envelope (SmallContent/NonBulkDense + CRC32C pair + firstRowNum) —
    the stream a real segment file would hold.  Returns bytes or None.
    nulls: optional per-row bool/uint8 null flags."""
    w = ref_writer()
    if w is None:
        return None
    vals = np.ascontiguousarray(vals)
    width = vals.itemsize
    cap = len(vals) * width + (1 << 21)
    out = np.zeros(cap, np.uint8)
    lens = np.zeros(65536, np.int32)
    rows = np.zeros(65536, np.int32)
    # header reserve: AoHeader_Size(isLong, checksum=true, firstRowNum=true)
    # = 24 for Orig (regular) streams, 32 for Dense (datumstream.c:588-605)
    reserve = 24 if version == 0 else 32
    nptr = 0
    if nulls is not None:
        nulls = np.ascontiguousarray(nulls, np.uint8)
        assert len(nulls) == len(vals)
        nptr = nulls.ctypes.data
    nb = w.refw_encode_nulls(vals.ctypes.data, nptr, width, len(vals),
                             version, rle, delta,
                             blocksize - reserve, out.ctypes.data, cap,
                             lens.ctypes.data, rows.ctypes.data, 65536)
    assert nb > 0, nb
    stream = bytearray()
    off = 0
    frn = 1
    for b in range(nb):
        content = out[off:off + lens[b]].tobytes()
        off += lens[b]
        logical = int(rows[b])
        clen = len(content)
        if logical <= 16383:
            b03 = (1 << 28) | (1 << 27) | (1 << 24) | (logical << 10) | (clen >> 11)
            b47 = (clen & 0x7FF) << 21
        else:
            b03 = (3 << 28) | (1 << 27) | (1 << 24) | (clen & 0x1FFFFF)
            b47 = logical & 0x3FFFFFFF
        blocklen = (24 + clen + 7) & ~7
        blk = bytearray(blocklen)
        blk[0:4] = b03.to_bytes(4, "little")
        blk[4:8] = b47.to_bytes(4, "little")
        blk[16:24] = frn.to_bytes(8, "little")
        blk[24:24 + clen] = content
        bc = lib.orc_crc32c(0xFFFFFFFF, bytes(blk[16:]), blocklen - 16)
        hc_in = bytes(blk[0:8]) + bc.to_bytes(4, "little")
        hc = lib.orc_crc32c(0xFFFFFFFF, hc_in, 12)
        blk[8:12] = bc.to_bytes(4, "little")
        blk[12:16] = hc.to_bytes(4, "little")
        stream += blk
        frn += logical
    return bytes(stream)


def set_threads(n):
    return lib.orc_set_threads(n)


def _np(ptr, n, dtype):
    return np.ctypeslib.as_array(ptr, shape=(n,)).astype(dtype, copy=True)


def gen_customer(sf, seed=42, seg=0, nsegs=1):
    t = _Customer()
    assert lib.orc_gen_customer(sf, seed, seg, nsegs, ctypes.byref(t)) == 0
    out = {"c_custkey": _np(t.c_custkey, t.n, np.int64),
           "c_mktsegment": _np(t.c_mktsegment, t.n, np.uint8)}
    lib.orc_free(t.c_custkey); lib.orc_free(t.c_mktsegment)
    return out


def gen_orders(sf, seed=42, seg=0, nsegs=1):
    t = _Orders()
    assert lib.orc_gen_orders(sf, seed, seg, nsegs, ctypes.byref(t)) == 0
    out = {"o_orderkey": _np(t.o_orderkey, t.n, np.int64),
           "o_custkey": _np(t.o_custkey, t.n, np.int64),
           "o_orderdate": _np(t.o_orderdate, t.n, np.int32),
           "o_shippriority": _np(t.o_shippriority, t.n, np.int32)}
    for f in ("o_orderkey", "o_custkey", "o_orderdate", "o_shippriority"):
        lib.orc_free(getattr(t, f))
    return out


def gen_lineitem(sf, seed=42, seg=0, nsegs=1):
    t = _Lineitem()
    assert lib.orc_gen_lineitem(sf, seed, seg, nsegs, ctypes.byref(t)) == 0
    out = {"l_orderkey": _np(t.l_orderkey, t.n, np.int64),
           "l_extendedprice": _np(t.l_extendedprice, t.n, np.float64),
           "l_discount": _np(t.l_discount, t.n, np.float64),
           "l_shipdate": _np(t.l_shipdate, t.n, np.int32)}
    for f in ("l_orderkey", "l_extendedprice", "l_discount", "l_shipdate"):
        lib.orc_free(getattr(t, f))
    return out


def _as_struct(table, st, fields):
    t = st()
    arrs = []
    for f, ct in fields:
        a = np.ascontiguousarray(table[f])
        arrs.append(a)
        setattr(t, f, a.ctypes.data_as(ctypes.POINTER(ct)))
    t.n = len(arrs[0])
    t._keepalive = arrs
    return t


def q3(cust, orders, lineitem, cutoff=None):
    """Run the oracle Q3 pipeline on numpy column dicts; returns a dict of arrays."""
    c = _as_struct(cust, _Customer, [("c_custkey", ctypes.c_int64),
                                     ("c_mktsegment", ctypes.c_uint8)])
    o = _as_struct(orders, _Orders, [("o_orderkey", ctypes.c_int64),
                                     ("o_custkey", ctypes.c_int64),
                                     ("o_orderdate", ctypes.c_int32),
                                     ("o_shippriority", ctypes.c_int32)])
    li = _as_struct(lineitem, _Lineitem, [("l_orderkey", ctypes.c_int64),
                                          ("l_extendedprice", ctypes.c_double),
                                          ("l_discount", ctypes.c_double),
                                          ("l_shipdate", ctypes.c_int32)])
    gp = ctypes.POINTER(_Group)()
    ng = lib.orc_q3(ctypes.byref(c), ctypes.byref(o), ctypes.byref(li),
                    CUTOFF_19950315 if cutoff is None else cutoff, ctypes.byref(gp))
    assert ng >= 0
    res = {"l_orderkey": np.array([gp[i].l_orderkey for i in range(ng)], np.int64),
           "o_orderdate": np.array([gp[i].o_orderdate for i in range(ng)], np.int32),
           "o_shippriority": np.array([gp[i].o_shippriority for i in range(ng)], np.int32),
           "revenue": np.array([gp[i].revenue for i in range(ng)], np.float64),
           "nitems": np.array([gp[i].nitems for i in range(ng)], np.int64)}
    lib.orc_free(gp)
    return res


def route(keys, nsegs):
    """Vectorised bit-exact Motion routing: jump_consistent_hash(cdbhash(k))."""
    keys = np.ascontiguousarray(keys, np.int64)
    out = np.zeros(len(keys), np.int32)
    lib.orc_route_i64_batch(keys.ctypes.data, len(keys), nsegs, out.ctypes.data)
    return out


def route_multi(vals, types, nsegs, isnull=None):
    """Multi-key Motion routing (cdbhash.c:189-247 rotate-combine): vals is
    (n, nkeys) int64 row-major; types[k] 0 = int8, 1 = int4/date."""
    vals = np.ascontiguousarray(vals, np.int64)
    n, nkeys = vals.shape
    types = np.ascontiguousarray(types, np.int32)
    nul = None
    nul_ptr = None
    if isnull is not None:
        nul = np.ascontiguousarray(isnull, np.uint8)
        assert nul.shape == vals.shape
        nul_ptr = nul.ctypes.data
    out = np.zeros(n, np.int32)
    lib.orc_route_multi_batch(vals.ctypes.data, nul_ptr, types.ctypes.data,
                              nkeys, n, nsegs, out.ctypes.data)
    return out


def cdbhash_multi(vals, types, isnull=None):
    vals = np.ascontiguousarray(vals, np.int64)
    types = np.ascontiguousarray(types, np.int32)
    nul_ptr = None
    if isnull is not None:
        nul = np.ascontiguousarray(isnull, np.uint8)
        nul_ptr = nul.ctypes.data
    return lib.orc_cdbhash_multi(vals.ctypes.data, nul_ptr,
                                 types.ctypes.data, len(vals))


def aocs_encode(vals):
    """Encode a fixed-width NOT NULL numpy column into an AOCS stream (bytes)."""
    vals = np.ascontiguousarray(vals)
    width = vals.itemsize
    n = len(vals)
    cap = lib.orc_aocs_encoded_size(width, n, 32768)
    buf = np.zeros(cap, np.uint8)
    got = lib.orc_aocs_encode(vals.ctypes.data, width, n, 1, 32768, buf.ctypes.data, cap)
    assert got == cap, (got, cap)
    return buf.tobytes()


def q3_numeric(cust, orders, lineitem, cutoff=None):
    """numeric(15,2) mode: exact scaled-int64 revenue numerators."""
    c = _as_struct(cust, _Customer, [("c_custkey", ctypes.c_int64),
                                     ("c_mktsegment", ctypes.c_uint8)])
    o = _as_struct(orders, _Orders, [("o_orderkey", ctypes.c_int64),
                                     ("o_custkey", ctypes.c_int64),
                                     ("o_orderdate", ctypes.c_int32),
                                     ("o_shippriority", ctypes.c_int32)])
    li = _as_struct(lineitem, _Lineitem, [("l_orderkey", ctypes.c_int64),
                                          ("l_extendedprice", ctypes.c_double),
                                          ("l_discount", ctypes.c_double),
                                          ("l_shipdate", ctypes.c_int32)])
    gp = ctypes.POINTER(_GroupN)()
    ng = lib.orc_q3_numeric(ctypes.byref(c), ctypes.byref(o), ctypes.byref(li),
                            CUTOFF_19950315 if cutoff is None else cutoff,
                            ctypes.byref(gp))
    assert ng >= 0
    res = {"l_orderkey": np.array([gp[i].l_orderkey for i in range(ng)], np.int64),
           "o_orderdate": np.array([gp[i].o_orderdate for i in range(ng)], np.int32),
           "o_shippriority": np.array([gp[i].o_shippriority for i in range(ng)], np.int32),
           "revenue_num": np.array([gp[i].revenue_num for i in range(ng)], np.int64),
           "nitems": np.array([gp[i].nitems for i in range(ng)], np.int64)}
    lib.orc_free(gp)
    return res


def aocs_encode_rle(vals):
    """RLE_TYPE (Dense_Enhanced) encode of a fixed-width NOT NULL column."""
    vals = np.ascontiguousarray(vals)
    width = vals.itemsize
    cap = len(vals) * width + (1 << 20)
    buf = np.zeros(cap, np.uint8)
    got = lib.orc_aocs_encode_rle(vals.ctypes.data, width, len(vals), 1, 32768,
                                  buf.ctypes.data, cap)
    assert got >= 0
    return buf[:got].tobytes()


def aocs_encode_rle_delta(vals):
    """Full Dense_Enhanced encode: RLE + DELTA_RANGE (sign-magnitude varint
    deltas ≤ 0x1FFFFFFF) for NOT NULL int columns."""
    vals = np.ascontiguousarray(vals)
    width = vals.itemsize
    cap = len(vals) * width + (1 << 20)
    buf = np.zeros(cap, np.uint8)
    got = lib.orc_aocs_encode_rle_delta(vals.ctypes.data, width, len(vals), 1,
                                        32768, buf.ctypes.data, cap)
    assert got >= 0
    return buf[:got].tobytes()


def aocs_encode_rle_delta_nulls(vals, nulls, delta=1):
    """Dense_Enhanced encode with a NULL bitmap (byte-exact vs reference)."""
    vals = np.ascontiguousarray(vals)
    nulls = np.ascontiguousarray(nulls, np.uint8)
    width = vals.itemsize
    cap = len(vals) * width + (1 << 20)
    buf = np.zeros(cap, np.uint8)
    got = lib.orc_aocs_encode_rle_delta_nulls(
        vals.ctypes.data, nulls.ctypes.data, width, len(vals), 1, 32768,
        delta, buf.ctypes.data, cap)
    assert got >= 0
    return buf[:got].tobytes()


def aocs_encode_orig_nulls(vals, nulls):
    """Original-version encode with a NULL bitmap (byte-exact vs reference)."""
    vals = np.ascontiguousarray(vals)
    nulls = np.ascontiguousarray(nulls, np.uint8)
    width = vals.itemsize
    cap = len(vals) * width + (1 << 21)
    buf = np.zeros(cap, np.uint8)
    got = lib.orc_aocs_encode_orig_nulls(
        vals.ctypes.data, nulls.ctypes.data, width, len(vals), 1, 32768,
        buf.ctypes.data, cap)
    assert got >= 0
    return buf[:got].tobytes()


def aocs_decode_nullable(stream, width, nrows, dtype, verify=1, codec=1):
    """Decode any AOCS stream incl. NULL-bearing blocks.
    Returns (values, validity) — null datums decode as zero."""
    out = np.zeros(nrows, dtype)
    validity = np.zeros(nrows, np.uint8)
    got = lib.orc_aocs_decode_nullable(stream, len(stream), width,
                                       out.ctypes.data, validity.ctypes.data,
                                       nrows, verify, codec)
    assert got == nrows, got
    return out, validity


def _varlena_args(strings, nulls):
    payload = b"".join(b"" if (nulls is not None and nulls[i]) else s
                       for i, s in enumerate(strings))
    offsets = np.zeros(len(strings) + 1, np.int64)
    w = 0
    for i, s in enumerate(strings):
        if nulls is None or not nulls[i]:
            w += len(s)
        offsets[i + 1] = w
    pay = np.frombuffer(payload, np.uint8).copy() if payload else np.zeros(1, np.uint8)
    nl = None
    nptr = 0
    if nulls is not None:
        nl = np.ascontiguousarray(nulls, np.uint8)
        nptr = nl.ctypes.data
    return pay, offsets, nl, nptr


def aocs_encode_varlena(strings, nulls=None, blocksize=32768):
    """Orig-format varlena (text) stream from a list of bytes values."""
    pay, offsets, nl, nptr = _varlena_args(strings, nulls)
    cap = int(offsets[-1]) + 16 * len(strings) + (1 << 20)
    buf = np.zeros(cap, np.uint8)
    got = lib.orc_aocs_encode_varlena(pay.ctypes.data, offsets.ctypes.data,
                                      nptr, len(strings), 1, blocksize,
                                      buf.ctypes.data, cap)
    assert got >= 0, got
    return buf[:got].tobytes()


def aocs_encode_varlena_rle(strings, nulls=None, blocksize=32768):
    """Dense_Enhanced rle_type varlena stream (RLE on repeated payloads)."""
    pay, offsets, nl, nptr = _varlena_args(strings, nulls)
    cap = int(offsets[-1]) + 16 * len(strings) + (1 << 20)
    buf = np.zeros(cap, np.uint8)
    got = lib.orc_aocs_encode_varlena_rle(pay.ctypes.data, offsets.ctypes.data,
                                          nptr, len(strings), 1, blocksize,
                                          buf.ctypes.data, cap)
    assert got >= 0, got
    return buf[:got].tobytes()


def aocs_decode_varlena(stream, nrows, verify=1):
    """Decode an Orig/Dense varlena stream -> (list of bytes|None).
    RLE streams can expand well past the stream size; grow on -3."""
    cap = len(stream) + 16
    while True:
        payload = np.zeros(cap, np.uint8)
        offsets = np.zeros(nrows + 1, np.int64)
        validity = np.zeros(nrows, np.uint8)
        got = lib.orc_aocs_decode_varlena(stream, len(stream), nrows,
                                          payload.ctypes.data, cap,
                                          offsets.ctypes.data,
                                          validity.ctypes.data, verify)
        if got == -3:
            cap *= 4
            continue
        break
    assert got == nrows, got
    out = []
    for i in range(nrows):
        if not validity[i]:
            out.append(None)
        else:
            out.append(payload[offsets[i]:offsets[i + 1]].tobytes())
    return out


def ref_writer_varlena_stream(strings, nulls=None, blocksize=32768,
                              version=0, rle=0):
    """REAL reference-writer varlena stream wrapped in our AO envelope."""
    w = ref_writer()
    if w is None:
        return None
    if not hasattr(w, "_varlena_decl"):
        w.refw_encode_varlena.restype = ctypes.c_int
        w.refw_encode_varlena.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
            ctypes.c_int, ctypes.c_int, ctypes.c_int32,
            ctypes.c_void_p, ctypes.c_int64,
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int]
        w._varlena_decl = True
    pay, offsets, nl, nptr = _varlena_args(strings, nulls)
    cap = int(offsets[-1]) + 16 * len(strings) + (1 << 21)
    out = np.zeros(cap, np.uint8)
    lens = np.zeros(65536, np.int32)
    rows = np.zeros(65536, np.int32)
    nb = w.refw_encode_varlena(pay.ctypes.data, offsets.ctypes.data, nptr,
                               len(strings), version, rle,
                               blocksize - (24 if version == 0 else 32),
                               out.ctypes.data, cap,
                               lens.ctypes.data, rows.ctypes.data, 65536)
    assert nb > 0, nb
    stream = bytearray()
    off = 0
    frn = 1
    for b in range(nb):
        content = out[off:off + lens[b]].tobytes()
        off += lens[b]
        logical = int(rows[b])
        clen = len(content)
        if logical <= 16383:
            b03 = (1 << 28) | (1 << 27) | (1 << 24) | (logical << 10) | (clen >> 11)
            b47 = (clen & 0x7FF) << 21
        else:
            b03 = (3 << 28) | (1 << 27) | (1 << 24) | (clen & 0x1FFFFF)
            b47 = logical & 0x3FFFFFFF
        blocklen = (24 + clen + 7) & ~7
        blk = bytearray(blocklen)
        blk[0:4] = b03.to_bytes(4, "little")
        blk[4:8] = b47.to_bytes(4, "little")
        blk[16:24] = frn.to_bytes(8, "little")
        blk[24:24 + clen] = content
        bc = lib.orc_crc32c(0xFFFFFFFF, bytes(blk[16:]), blocklen - 16)
        hc = lib.orc_crc32c(0xFFFFFFFF, bytes(blk[0:8]) + bc.to_bytes(4, "little"), 12)
        blk[8:12] = bc.to_bytes(4, "little")
        blk[12:16] = hc.to_bytes(4, "little")
        stream += blk
        frn += logical
    return bytes(stream)


def aocs_encode_zlib(vals, level=6):
    """zlib bulk-compressed AOCS stream (compresstype=zlib)."""
    vals = np.ascontiguousarray(vals)
    width = vals.itemsize
    cap = len(vals) * width + (1 << 20)
    buf = np.zeros(cap, np.uint8)
    got = lib.orc_aocs_encode_zlib(vals.ctypes.data, width, len(vals), 1,
                                   32768, level, buf.ctypes.data, cap)
    assert got >= 0
    return buf[:got].tobytes()


def aocs_encode_zstd(vals, level=3):
    """zstd bulk-compressed AOCS stream (compresstype=zstd, gpcontrib/zstd)."""
    vals = np.ascontiguousarray(vals)
    width = vals.itemsize
    cap = len(vals) * width + (1 << 20)
    buf = np.zeros(cap, np.uint8)
    got = lib.orc_aocs_encode_zstd(vals.ctypes.data, width, len(vals), 1,
                                   32768, level, buf.ctypes.data, cap)
    assert got >= 0
    return buf[:got].tobytes()


def aocs_decode(stream, width, nrows, dtype, verify=True):
    buf = np.frombuffer(stream, np.uint8)
    out = np.zeros(nrows, dtype)
    got = lib.orc_aocs_decode(buf.ctypes.data, len(buf), width,
                              out.ctypes.data, nrows, 1 if verify else 0)
    assert got == nrows, got
    return out
