"""cloudberry_amd — Python plumbing around libgpuexec.so (the C-ABI product
path; see include/gpuexec.h).  This wrapper exists for tests and the bench
harness; the library itself has no Python or torch dependency.

The GPU path NEVER falls back to CPU: if the HIP library is missing or no
device is usable, every entry point raises."""
import ctypes
import os

import numpy as np

_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_SO = os.path.join(_ROOT, "cloudberry_amd", "libgpuexec.so")

TPCH_CUSTOMER, TPCH_ORDERS, TPCH_LINEITEM = 0, 1, 2
TPCH_LINEITEM_NUMERIC = 3
TPCH_LINEITEM_RLEKEY = 4
TPCH_LINEITEM_Q1 = 5
CUTOFF_19950315 = -1753  # DateADT of 1995-03-15 (validated vs oracle in tests)

_STATUS = {0: "GX_OK", 1: "GX_ERR_HIP", 2: "GX_ERR_RCCL", 3: "GX_ERR_INVALID",
           4: "GX_ERR_CHECKSUM", 5: "GX_ERR_OOM", 6: "GX_ERR_NOGPU", 7: "GX_ERR_STATE"}


class GxError(RuntimeError):
    def __init__(self, status, detail=""):
        self.status = status
        super().__init__(f"{_STATUS.get(status, status)}: {detail}")


class _Group(ctypes.Structure):
    _fields_ = [("l_orderkey", ctypes.c_int64),
                ("o_orderdate", ctypes.c_int32),
                ("o_shippriority", ctypes.c_int32),
                ("revenue", ctypes.c_double),
                ("revenue_num", ctypes.c_int64),
                ("nitems", ctypes.c_int64),
                ("key_is_null", ctypes.c_uint8),
                ("attrs_null", ctypes.c_uint8),
                ("_pad", ctypes.c_uint8 * 6)]


class _Stats(ctypes.Structure):
    _fields_ = [("ms_cust_build", ctypes.c_double),
                ("ms_orders_build", ctypes.c_double),
                ("ms_probe_agg", ctypes.c_double),
                ("ms_extract", ctypes.c_double),
                ("ms_motion", ctypes.c_double),
                ("ms_total", ctypes.c_double),
                ("cust_rows", ctypes.c_int64),
                ("ord_rows", ctypes.c_int64),
                ("li_rows", ctypes.c_int64),
                ("probe_hits", ctypes.c_int64),
                ("groups", ctypes.c_int64),
                ("bytes_scanned", ctypes.c_double),
                ("ms_motion_counts", ctypes.c_double),
                ("ms_motion_payload", ctypes.c_double)]


class _KvGroup(ctypes.Structure):
    _fields_ = [("key", ctypes.c_int64), ("key_is_null", ctypes.c_uint8),
                ("_pad", ctypes.c_uint8 * 7), ("sum", ctypes.c_double),
                ("count", ctypes.c_int64)]


class _Filter(ctypes.Structure):
    _fields_ = [("col", ctypes.c_int32), ("op", ctypes.c_int32),
                ("literal", ctypes.c_int64)]


class _Q3Desc(ctypes.Structure):
    _fields_ = [("dim", ctypes.c_void_p), ("dim_key_col", ctypes.c_int32),
                ("dim_filter", _Filter),
                ("mid", ctypes.c_void_p), ("mid_key_col", ctypes.c_int32),
                ("mid_fk_col", ctypes.c_int32), ("mid_attr1_col", ctypes.c_int32),
                ("mid_attr2_col", ctypes.c_int32), ("mid_filter", _Filter),
                ("fact", ctypes.c_void_p), ("fact_key_col", ctypes.c_int32),
                ("fact_a_col", ctypes.c_int32), ("fact_b_col", ctypes.c_int32),
                ("fact_filter", _Filter),
                ("dim_text", ctypes.c_char * 64),
                ("dim_text_len", ctypes.c_int32),
                ("dim_extra", _Filter * 4), ("mid_extra", _Filter * 4),
                ("fact_extra", _Filter * 4),
                ("n_dim_extra", ctypes.c_int32),
                ("n_mid_extra", ctypes.c_int32),
                ("n_fact_extra", ctypes.c_int32),
                ("dim_join", ctypes.c_int32),
                ("fact_join", ctypes.c_int32)]


class _ColDesc(ctypes.Structure):
    _fields_ = [("host_stream", ctypes.c_void_p),
                ("nbytes", ctypes.c_int64),
                ("width", ctypes.c_int32),
                ("nrows", ctypes.c_int64),
                ("blocksize", ctypes.c_int32),
                ("format", ctypes.c_int32),
                ("codec", ctypes.c_int32)]


def _load():
    if not os.path.exists(_SO):
        raise RuntimeError(
            f"HIP extension missing: {_SO}. Build it with "
            f"`python -c \"import __graft_entry__; __graft_entry__.build()\"` — "
            f"the GPU executor has no CPU fallback.")
    lib = ctypes.CDLL(_SO)
    lib.gx_last_error.restype = ctypes.c_char_p
    lib.gx_last_error.argtypes = [ctypes.c_void_p]
    lib.gx_version.restype = ctypes.c_char_p
    lib.gx_init.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_int,
                            ctypes.POINTER(ctypes.c_void_p)]
    lib.gx_shutdown.argtypes = [ctypes.c_void_p]
    lib.gx_comm_unique_id.argtypes = [ctypes.c_char_p]
    lib.gx_comm_init.argtypes = [ctypes.c_void_p, ctypes.c_char_p]
    lib.gx_table_bind.argtypes = [ctypes.c_void_p, ctypes.POINTER(_ColDesc),
                                  ctypes.c_int, ctypes.POINTER(ctypes.c_void_p)]
    lib.gx_table_free.argtypes = [ctypes.c_void_p]
    lib.gx_table_nrows.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_int64)]
    lib.gx_table_logical_bytes.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_double)]
    lib.gx_tpch_gen.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_double,
                                ctypes.c_uint64, ctypes.POINTER(ctypes.c_void_p)]
    lib.gx_table_dump_stream.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                         ctypes.c_int, ctypes.c_void_p,
                                         ctypes.c_int64,
                                         ctypes.POINTER(ctypes.c_int64)]
    lib.gx_decode_column.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int,
                                     ctypes.c_void_p, ctypes.c_int64, ctypes.c_int]
    lib.gx_decode_column_nullable.restype = ctypes.c_int
    lib.gx_decode_column_nullable.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_int]
    lib.gx_table_set_visimap.restype = ctypes.c_int
    lib.gx_table_set_visimap.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                         ctypes.c_void_p, ctypes.c_int64]
    lib.gx_decode_column_varlena.restype = ctypes.c_int
    lib.gx_decode_column_varlena.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
        ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p, ctypes.c_int]
    lib.gx_q1.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int32,
                          ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                          ctypes.POINTER(ctypes.c_double)]
    lib.gx_scan_filter.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int,
                                   ctypes.c_int, ctypes.c_int64,
                                   ctypes.POINTER(ctypes.c_int64),
                                   ctypes.POINTER(ctypes.c_double)]
    lib.gx_partition.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
                                 ctypes.c_int32, ctypes.c_void_p]
    lib.gx_groupby.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int,
                               ctypes.c_int,
                               ctypes.POINTER(ctypes.POINTER(_KvGroup)),
                               ctypes.POINTER(ctypes.c_int64)]
    lib.gx_partition_multi.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.c_int32, ctypes.c_int64,
                                       ctypes.c_int32, ctypes.c_void_p]
    lib.gx_q3_prepare.argtypes = [ctypes.c_void_p] * 4 + [ctypes.c_int32,
                                  ctypes.POINTER(ctypes.c_void_p)]
    lib.gx_q3_prepare_desc.argtypes = [ctypes.c_void_p, ctypes.POINTER(_Q3Desc),
                                       ctypes.POINTER(ctypes.c_void_p)]
    lib.gx_q3_set_numeric.argtypes = [ctypes.c_void_p, ctypes.c_int]
    lib.gx_q3_run.argtypes = [ctypes.c_void_p]
    lib.gx_q3_stats_get.argtypes = [ctypes.c_void_p, ctypes.POINTER(_Stats)]
    lib.gx_q3_result.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.POINTER(_Group)),
                                 ctypes.POINTER(ctypes.c_int64)]
    lib.gx_q3_topn.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
                               ctypes.POINTER(ctypes.c_int64)]
    lib.gx_q3_free.argtypes = [ctypes.c_void_p]
    lib.gx_free.argtypes = [ctypes.c_void_p]
    lib.gx_test_motion1.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int32,
                                    ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p,
                                    ctypes.c_int64, ctypes.POINTER(ctypes.c_int64)]
    lib.gx_test_qual.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
                                 ctypes.c_int64, ctypes.c_int, ctypes.c_void_p,
                                 ctypes.c_void_p, ctypes.c_int64,
                                 ctypes.POINTER(ctypes.c_int64)]
    lib.gx_test_q3_from_qual.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                         ctypes.c_int64, ctypes.c_void_p,
                                         ctypes.c_int32,
                                         ctypes.POINTER(ctypes.POINTER(_Group)),
                                         ctypes.POINTER(ctypes.c_int64)]
    return lib


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = _load()
    return _lib


class Context:
    def __init__(self, device=0, seg=0, nsegs=1):
        self._lib = lib()
        self._h = ctypes.c_void_p()
        self._chk(self._lib.gx_init(device, seg, nsegs, ctypes.byref(self._h)), None)
        self.seg, self.nsegs = seg, nsegs

    def _chk(self, st, h="self"):
        if st != 0:
            handle = self._h if h == "self" else h
            msg = self._lib.gx_last_error(handle)
            raise GxError(st, (msg or b"").decode())

    def comm_unique_id(self):
        buf = ctypes.create_string_buffer(128)
        self._chk(self._lib.gx_comm_unique_id(buf), None)
        return buf.raw

    def comm_init(self, uid: bytes):
        assert len(uid) == 128
        self._chk(self._lib.gx_comm_init(self._h, uid))

    def tpch_gen(self, which, sf, seed=42):
        t = ctypes.c_void_p()
        self._chk(self._lib.gx_tpch_gen(self._h, which, sf, seed, ctypes.byref(t)))
        return Table(self, t)

    def bind(self, streams):
        """streams: list of (bytes, width, nrows[, format]) AOCS column
        streams; format 0 = Orig fixed, 1 = Dense/RLE."""
        descs = (_ColDesc * len(streams))()
        keep = []
        for i, spec in enumerate(streams):
            data, width, nrows = spec[:3]
            arr = np.frombuffer(data, np.uint8)
            keep.append(arr)
            descs[i].host_stream = arr.ctypes.data
            descs[i].nbytes = len(arr)
            descs[i].width = width
            descs[i].nrows = nrows
            descs[i].blocksize = 32768
            descs[i].format = spec[3] if len(spec) > 3 else 0
            descs[i].codec = spec[4] if len(spec) > 4 else 0
        t = ctypes.c_void_p()
        self._chk(self._lib.gx_table_bind(self._h, descs, len(streams), ctypes.byref(t)))
        tb = Table(self, t)
        tb._col_nbytes = [len(s[0]) for s in streams]
        return tb

    def partition(self, keys, nsegs):
        keys = np.ascontiguousarray(keys, np.int64)
        out = np.zeros(len(keys), np.int32)
        self._chk(self._lib.gx_partition(self._h, keys.ctypes.data, len(keys),
                                         nsegs, out.ctypes.data))
        return out

    def partition_multi(self, vals, types, nsegs, isnull=None):
        """Multi-column distribution-key routing (cdbhash rotate-combine);
        vals (n, nkeys) int64; types[k] 0 = int8, 1 = int4/date."""
        vals = np.ascontiguousarray(vals, np.int64)
        n, nkeys = vals.shape
        types = np.ascontiguousarray(types, np.int32)
        nul_ptr = None
        if isnull is not None:
            nul = np.ascontiguousarray(isnull, np.uint8)
            assert nul.shape == vals.shape
            nul_ptr = nul.ctypes.data
        out = np.zeros(n, np.int32)
        self._chk(self._lib.gx_partition_multi(
            self._h, vals.ctypes.data, nul_ptr, types.ctypes.data,
            nkeys, n, nsegs, out.ctypes.data))
        return out

    def groupby(self, table, key_col, val_col):
        """GROUP BY key_col with COUNT(*)/SUM(val_col); NULL keys form one
        group (returned last, key_is_null True)."""
        gp = ctypes.POINTER(_KvGroup)()
        n = ctypes.c_int64()
        self._chk(self._lib.gx_groupby(self._h, table._t, key_col, val_col,
                                       ctypes.byref(gp), ctypes.byref(n)))
        n = n.value
        res = {"key": np.array([gp[i].key for i in range(n)], np.int64),
               "key_is_null": np.array([gp[i].key_is_null for i in range(n)],
                                       np.bool_),
               "sum": np.array([gp[i].sum for i in range(n)], np.float64),
               "count": np.array([gp[i].count for i in range(n)], np.int64)}
        self._lib.gx_free(gp)
        return res

    def test_motion1(self, orders, nsegs, cutoff=CUTOFF_19950315):
        """Run the Motion-1 partition kernels; returns (counts, rows dict)."""
        cap = orders.nrows
        counts = np.zeros(nsegs, np.int64)
        rows = np.zeros(cap, dtype=[("okey", np.int64), ("ocust", np.int64),
                                    ("odate", np.int32), ("oprio", np.int32)])
        total = ctypes.c_int64()
        self._chk(self._lib.gx_test_motion1(self._h, orders._t, cutoff, nsegs,
                                            counts.ctypes.data, rows.ctypes.data,
                                            cap, ctypes.byref(total)))
        return counts, rows[:total.value]

    QUAL_DTYPE = np.dtype([("okey", np.int64), ("odate", np.int32),
                           ("oprio", np.int32)])

    def test_qual(self, customer, rows, nsegs, ):
        """Motion stage 2 on one GPU: semijoin received orders rows against
        this segment's customer set, route qualifiers by o_orderkey."""
        counts = np.zeros(nsegs, np.int64)
        out = np.zeros(len(rows), self.QUAL_DTYPE)
        total = ctypes.c_int64()
        self._chk(self._lib.gx_test_qual(self._h, customer._t, rows.ctypes.data,
                                         len(rows), nsegs, counts.ctypes.data,
                                         out.ctypes.data, len(rows),
                                         ctypes.byref(total)))
        return counts, out[:total.value]

    def test_q3_from_qual(self, qual_rows, lineitem, cutoff=CUTOFF_19950315):
        """Motion stage 3 on one GPU: build table from qual rows, probe the
        local lineitem, return groups sorted by key."""
        gp = ctypes.POINTER(_Group)()
        n = ctypes.c_int64()
        self._chk(self._lib.gx_test_q3_from_qual(self._h, qual_rows.ctypes.data,
                                                 len(qual_rows), lineitem._t,
                                                 cutoff, ctypes.byref(gp),
                                                 ctypes.byref(n)))
        n = n.value
        res = {"l_orderkey": np.array([gp[i].l_orderkey for i in range(n)], np.int64),
               "o_orderdate": np.array([gp[i].o_orderdate for i in range(n)], np.int32),
               "o_shippriority": np.array([gp[i].o_shippriority for i in range(n)], np.int32),
               "revenue": np.array([gp[i].revenue for i in range(n)], np.float64),
               "nitems": np.array([gp[i].nitems for i in range(n)], np.int64),
               "key_is_null": np.array([gp[i].key_is_null for i in range(n)],
                                       np.bool_),
               "attrs_null": np.array([gp[i].attrs_null for i in range(n)],
                                      np.bool_)}
        self._lib.gx_free(gp)
        return res

    def q1(self, lineitem_q1, cutoff):
        """TPC-H Q1 core: returns dict of 6-group arrays + kernel ms."""
        counts = np.zeros(6, np.int64)
        sp = np.zeros(6, np.float64)
        sr = np.zeros(6, np.float64)
        ms = ctypes.c_double()
        self._chk(self._lib.gx_q1(self._h, lineitem_q1._t, cutoff,
                                  counts.ctypes.data, sp.ctypes.data,
                                  sr.ctypes.data, ctypes.byref(ms)))
        return {"count": counts, "sum_price": sp, "sum_revenue": sr,
                "avg_price": np.divide(sp, counts, out=np.zeros(6),
                                       where=counts > 0),
                "ms": ms.value}

    def q3_desc(self, desc_dict):
        """Plan-descriptor Q3 slice: column roles + filter
        {col, op, literal} triples chosen by the caller."""
        ops = {"<": 0, ">": 1, "==": 2, "!=": 3, "<=": 4, ">=": 5}
        d = _Q3Desc()
        for role in ("dim", "mid", "fact"):
            setattr(d, role, desc_dict[role]._t)
        d.dim_key_col = desc_dict["dim_key_col"]
        d.mid_key_col = desc_dict["mid_key_col"]
        d.mid_fk_col = desc_dict["mid_fk_col"]
        d.mid_attr1_col = desc_dict["mid_attr1_col"]
        d.mid_attr2_col = desc_dict["mid_attr2_col"]
        d.fact_key_col = desc_dict["fact_key_col"]
        d.fact_a_col = desc_dict["fact_a_col"]
        d.fact_b_col = desc_dict["fact_b_col"]
        for role in ("dim_filter", "mid_filter", "fact_filter"):
            col, op, lit = desc_dict[role]
            if isinstance(lit, (str, bytes)):
                # TEXT predicate (texteq on a varlena column, dim only)
                assert role == "dim_filter" and op == "=="
                blit = lit.encode() if isinstance(lit, str) else lit
                d.dim_text = blit
                d.dim_text_len = len(blit)
                setattr(d, role, _Filter(col, ops[op], 0))
            else:
                setattr(d, role, _Filter(col, ops[op], int(lit)))
        # AND-ed extra qual lists (execScan.c:241 semantics)
        for role, arr, cnt in (("dim_extra", d.dim_extra, "n_dim_extra"),
                               ("mid_extra", d.mid_extra, "n_mid_extra"),
                               ("fact_extra", d.fact_extra, "n_fact_extra")):
            quals = desc_dict.get(role, [])
            assert len(quals) <= 4
            for i, (col, op, lit) in enumerate(quals):
                arr[i] = _Filter(col, ops[op], int(lit))
            setattr(d, cnt, len(quals))
        d.dim_join = {"semi": 0, "anti": 1, "anti_notin": 2}[
            desc_dict.get("dim_join", "semi")]
        d.fact_join = {"inner": 0, "left_outer": 1}[
            desc_dict.get("fact_join", "inner")]
        q = ctypes.c_void_p()
        self._chk(self._lib.gx_q3_prepare_desc(self._h, ctypes.byref(d),
                                               ctypes.byref(q)))
        return Q3(self, q,
                  tables=(desc_dict["dim"], desc_dict["mid"], desc_dict["fact"]))

    def q3(self, cust, orders, lineitem, cutoff=CUTOFF_19950315, numeric=False):
        q = ctypes.c_void_p()
        self._chk(self._lib.gx_q3_prepare(self._h, cust._t, orders._t, lineitem._t,
                                          cutoff, ctypes.byref(q)))
        if numeric:
            self._chk(self._lib.gx_q3_set_numeric(q, 1))
        return Q3(self, q, tables=(cust, orders, lineitem))

    def close(self):
        if self._h:
            self._lib.gx_shutdown(self._h)
            self._h = None


class Table:
    def __init__(self, ctx, t):
        self.ctx = ctx
        self._t = t

    @property
    def nrows(self):
        n = ctypes.c_int64()
        self.ctx._chk(self.ctx._lib.gx_table_nrows(self._t, ctypes.byref(n)))
        return n.value

    @property
    def logical_bytes(self):
        b = ctypes.c_double()
        self.ctx._chk(self.ctx._lib.gx_table_logical_bytes(self._t, ctypes.byref(b)))
        return b.value

    def scan_filter(self, col, op, literal):
        """SeqScan+qual count; op in {'<','>','==','!='}; returns (count, ms)."""
        opc = {"<": 0, ">": 1, "==": 2, "!=": 3}[op]
        n = ctypes.c_int64()
        ms = ctypes.c_double()
        self.ctx._chk(self.ctx._lib.gx_scan_filter(self.ctx._h, self._t, col, opc,
                                                   literal, ctypes.byref(n),
                                                   ctypes.byref(ms)))
        return n.value, ms.value

    def dump_stream(self, col):
        """Raw AOCS stream bytes of a column (byte-level parity tests)."""
        cap = self.nrows * 16 + (1 << 22)
        buf = np.zeros(cap, np.uint8)
        n = ctypes.c_int64()
        self.ctx._chk(self.ctx._lib.gx_table_dump_stream(
            self.ctx._h, self._t, col, buf.ctypes.data, cap, ctypes.byref(n)))
        return buf[:n.value].tobytes()

    def decode_column(self, col, dtype, verify=True):
        n = self.nrows
        out = np.zeros(n, dtype)
        self.ctx._chk(self.ctx._lib.gx_decode_column(
            self.ctx._h, self._t, col, out.ctypes.data, n, 1 if verify else 0))
        return out

    def _coldesc_nbytes(self, col):
        return self._col_nbytes[col] if hasattr(self, "_col_nbytes") else 1 << 22

    def decode_column_varlena(self, col, verify=True):
        """Decode a varlena (text) directory column -> list of bytes|None.
        RLE streams expand past the stream size; grow the buffer on demand."""
        n = self.nrows
        cap = self._coldesc_nbytes(col) + 16
        while True:
            offsets = np.zeros(n + 1, np.int64)
            payload = np.zeros(cap, np.uint8)
            validity = np.zeros(max(n, 1), np.uint8)
            try:
                self.ctx._chk(self.ctx._lib.gx_decode_column_varlena(
                    self.ctx._h, self._t, col, offsets.ctypes.data,
                    payload.ctypes.data, cap, validity.ctypes.data,
                    1 if verify else 0))
            except GxError as e:
                if "too small" in str(e):
                    cap *= 4
                    continue
                raise
            break
        return [None if not validity[i]
                else payload[offsets[i]:offsets[i + 1]].tobytes()
                for i in range(n)]

    def decode_column_nullable(self, col, dtype, verify=True):
        """Decode a (possibly NULL-bearing) block-directory column.
        Returns (values, validity) — null datums decode as zero."""
        n = self.nrows
        out = np.zeros(n, dtype)
        validity = np.zeros(n, np.uint8)
        self.ctx._chk(self.ctx._lib.gx_decode_column_nullable(
            self.ctx._h, self._t, col, out.ctypes.data, validity.ctypes.data,
            n, 1 if verify else 0))
        return out, validity

    def free(self):
        if self._t:
            self.ctx._lib.gx_table_free(self._t)
            self._t = None

    def set_visimap(self, deleted):
        """Attach an AO visimap: `deleted` is a per-row bool array
        (True = tuple hidden/deleted); None clears.  Set before ctx.q3()."""
        if deleted is None:
            self.ctx._chk(self.ctx._lib.gx_table_set_visimap(
                self.ctx._h, self._t, None, 0))
            self._vmap = None
            return
        deleted = np.ascontiguousarray(deleted, np.uint8)
        assert len(deleted) == self.nrows
        bits = np.packbits(deleted, bitorder="little")
        self.ctx._chk(self.ctx._lib.gx_table_set_visimap(
            self.ctx._h, self._t, bits.ctypes.data, len(deleted)))
        self._vmap = bits        # keep the host copy alive until replaced

    def __del__(self):
        try:
            self.free()
        except Exception:
            pass            # interpreter teardown / context already gone


class Q3:
    def __init__(self, ctx, q, tables=()):
        self.ctx = ctx
        self._q = q
        self._tables = tables    # keep the Tables alive: the native Q3
                                 # holds raw device pointers into them

    def run(self):
        self.ctx._chk(self.ctx._lib.gx_q3_run(self._q))
        return self

    def stats(self):
        s = _Stats()
        self.ctx._chk(self.ctx._lib.gx_q3_stats_get(self._q, ctypes.byref(s)))
        return {f: getattr(s, f) for f, _ in s._fields_}

    def result(self):
        gp = ctypes.POINTER(_Group)()
        n = ctypes.c_int64()
        self.ctx._chk(self.ctx._lib.gx_q3_result(self._q, ctypes.byref(gp),
                                                 ctypes.byref(n)))
        n = n.value
        res = {"l_orderkey": np.array([gp[i].l_orderkey for i in range(n)], np.int64),
               "o_orderdate": np.array([gp[i].o_orderdate for i in range(n)], np.int32),
               "o_shippriority": np.array([gp[i].o_shippriority for i in range(n)], np.int32),
               "revenue": np.array([gp[i].revenue for i in range(n)], np.float64),
               "revenue_num": np.array([gp[i].revenue_num for i in range(n)], np.int64),
               "nitems": np.array([gp[i].nitems for i in range(n)], np.int64),
               "key_is_null": np.array([gp[i].key_is_null for i in range(n)],
                                       np.bool_),
               "attrs_null": np.array([gp[i].attrs_null for i in range(n)],
                                      np.bool_)}
        self.ctx._lib.gx_free(gp)
        return res

    def topn(self, n=10):
        out = (_Group * n)()
        m = ctypes.c_int64()
        self.ctx._chk(self.ctx._lib.gx_q3_topn(self._q, n, out, ctypes.byref(m)))
        m = m.value
        return {"l_orderkey": np.array([out[i].l_orderkey for i in range(m)], np.int64),
                "o_orderdate": np.array([out[i].o_orderdate for i in range(m)], np.int32),
                "o_shippriority": np.array([out[i].o_shippriority for i in range(m)], np.int32),
                "revenue": np.array([out[i].revenue for i in range(m)], np.float64),
                "revenue_num": np.array([out[i].revenue_num for i in range(m)], np.int64),
                "nitems": np.array([out[i].nitems for i in range(m)], np.int64),
                "key_is_null": np.array([out[i].key_is_null for i in range(m)],
                                        np.bool_),
                "attrs_null": np.array([out[i].attrs_null for i in range(m)],
                                       np.bool_)}

    def free(self):
        if self._q:
            self.ctx._lib.gx_q3_free(self._q)
            self._q = None

    def __del__(self):
        try:
            self.free()
        except Exception:
            pass

