"""GPU parity tests (pytest -m gpu, run on a real MI355X): the HIP path
against the oracle on identical seeded inputs, plus size-independent
properties at larger sizes.  All calls go through the C-ABI."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu

gx = pytest.importorskip("cloudberry_amd")


@pytest.fixture(scope="module")
def ctx():
    c = gx.Context(device=0, seg=0, nsegs=1)
    yield c
    c.close()


@pytest.fixture(scope="module")
def orc():
    from oracle import pyapi
    return pyapi


# ---------------- Motion routing: bit-exact ----------------

def test_partition_bit_exact(ctx, orc):
    rng = np.random.default_rng(5)
    keys = np.concatenate([np.arange(1, 1000, dtype=np.int64),
                           rng.integers(-2**62, 2**62, 20000).astype(np.int64)])
    for nsegs in (2, 3, 8, 64):
        got = ctx.partition(keys, nsegs)
        want = orc.route(keys, nsegs)
        assert (got == want).all(), f"routing diverges at nsegs={nsegs}"


# ---------------- AOCS codec: byte-identical streams ----------------

def test_gpu_encode_matches_oracle_encoder_bytes(ctx, orc):
    """gx_tpch_gen encodes on device; decoding through the C-ABI must match
    the oracle's values, and a table bound FROM oracle-encoded host streams
    must decode identically (checksums verified on device)."""
    sf = 0.01
    t = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    want = orc.gen_orders(sf)
    assert t.nrows == len(want["o_orderkey"])
    np.testing.assert_array_equal(t.decode_column(0, np.int64), want["o_orderkey"])
    np.testing.assert_array_equal(t.decode_column(1, np.int64), want["o_custkey"])
    np.testing.assert_array_equal(t.decode_column(2, np.int32), want["o_orderdate"])
    np.testing.assert_array_equal(t.decode_column(3, np.int32), want["o_shippriority"])
    t.free()


def test_bind_oracle_streams_roundtrip(ctx, orc):
    rng = np.random.default_rng(9)
    vals8 = rng.integers(-2**60, 2**60, 12345, dtype=np.int64)
    vals4 = rng.integers(-2**30, 2**30, 12345, dtype=np.int32)
    t = ctx.bind([(orc.aocs_encode(vals8), 8, len(vals8)),
                  (orc.aocs_encode(vals4), 4, len(vals4))])
    np.testing.assert_array_equal(t.decode_column(0, np.int64, verify=True), vals8)
    np.testing.assert_array_equal(t.decode_column(1, np.int32, verify=True), vals4)
    t.free()


def test_rle_stream_gpu_decode(ctx, orc):
    """Dense_Enhanced RLE streams bound through the C-ABI decode on device to
    the original values (clustered keys like l_orderkey), checksums verified."""
    rng = np.random.default_rng(12)
    keys = np.repeat(np.arange(1, 60000, dtype=np.int64),
                     rng.integers(1, 8, 59999))
    n = len(keys)
    dates = rng.integers(-3000, 0, n).astype(np.int32)   # incompressible
    t = ctx.bind([(orc.aocs_encode_rle(keys), 8, n, 1),
                  (orc.aocs_encode_rle(dates), 4, n, 1)])
    np.testing.assert_array_equal(t.decode_column(0, np.int64, verify=True), keys)
    np.testing.assert_array_equal(t.decode_column(1, np.int32, verify=True), dates)
    t.free()


def test_rle_q3_input_rejected(ctx, orc):
    """RLE on key/filter roles is materialized at prepare (r2), but a
    Dense/RLE MEASURE column (fact price) must still be rejected loudly,
    never silently mis-scanned."""
    c1 = ctx.tpch_gen(gx.TPCH_CUSTOMER, 0.01)
    c2 = ctx.tpch_gen(gx.TPCH_ORDERS, 0.01)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, 0.01)
    n = li.nrows
    keys = li.decode_column(0, np.int64)
    ship = li.decode_column(3, np.int32)
    price_rle = np.repeat(np.float64(7.0), n)   # compressible measure
    li_bad = ctx.bind([(orc.aocs_encode(keys), 8, n),
                       (orc.aocs_encode_rle(price_rle.view(np.int64)), 8, n, 1),
                       (orc.aocs_encode(np.zeros(n)), 8, n),
                       (orc.aocs_encode(ship), 4, n)])
    with pytest.raises(gx.GxError):
        ctx.q3(c1, c2, li_bad)
    li_bad.free(); li.free(); c2.free(); c1.free()


def test_corrupted_stream_rejected(ctx, orc):
    vals = np.arange(5000, dtype=np.int64)
    s = bytearray(orc.aocs_encode(vals))
    s[100] ^= 0x1          # flip a datum byte in block 0
    t = ctx.bind([(bytes(s), 8, len(vals))])
    with pytest.raises(gx.GxError) as ei:
        t.decode_column(0, np.int64, verify=True)
    assert ei.value.status == 4  # GX_ERR_CHECKSUM
    t.free()


# ---------------- Q3 parity vs oracle (small sizes) ----------------

def _q3_parity_at(ctx, orc, sf, seed=42):
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf, seed)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf, seed)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf, seed)
    q = ctx.q3(cust, ordr, li).run()
    got = q.result()
    st = q.stats()
    want = orc.q3(orc.gen_customer(sf, seed), orc.gen_orders(sf, seed),
                  orc.gen_lineitem(sf, seed))
    # bit-exact: group keys, dates, priorities, counts (integer work)
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["o_orderdate"], want["o_orderdate"])
    np.testing.assert_array_equal(got["o_shippriority"], want["o_shippriority"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    # f64 SUM within 1e-6 relative (BASELINE tolerance; ordering differs)
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-6)
    assert st["probe_hits"] == int(want["nitems"].sum())
    q.free(); li.free(); ordr.free(); cust.free()
    return len(want["l_orderkey"])


def test_q3_parity_sf001(ctx, orc):
    ng = _q3_parity_at(ctx, orc, 0.01)
    assert ng > 50


def test_q3_parity_sf01(ctx, orc):
    ng = _q3_parity_at(ctx, orc, 0.1)
    assert ng > 500


def test_q3_parity_sf1(ctx, orc):
    ng = _q3_parity_at(ctx, orc, 1.0)
    assert ng > 5000


def test_q3_parity_other_seed(ctx, orc):
    _q3_parity_at(ctx, orc, 0.05, seed=7)


def _top10(res):
    """Q3's final ORDER BY revenue DESC, o_orderdate LIMIT 10 (runs on the
    QD in the reference — trivial rows, host-side here)."""
    order = np.lexsort((res["o_orderdate"], -res["revenue"]))[:10]
    return {k: v[order] for k, v in res.items()}


def test_q3_top10_matches_oracle(ctx, orc):
    sf = 0.5
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    q = ctx.q3(cust, ordr, li).run()
    got = _top10(q.result())
    want = _top10(orc.q3(orc.gen_customer(sf), orc.gen_orders(sf),
                         orc.gen_lineitem(sf)))
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-6)
    # the DEVICE top-N kernel (gx_q3_topn) must agree with the host sort
    dev = q.topn(10)
    np.testing.assert_array_equal(dev["l_orderkey"], got["l_orderkey"])
    np.testing.assert_array_equal(dev["o_orderdate"], got["o_orderdate"])
    np.testing.assert_array_equal(dev["nitems"], got["nitems"])
    np.testing.assert_allclose(dev["revenue"], got["revenue"], rtol=0)
    q.free(); li.free(); ordr.free(); cust.free()


# ---------------- Motion partition kernels (one GPU) ----------------

def test_motion1_partition_kernels(ctx, orc):
    """k_ord_m1_hist/emit: every emitted row routed bit-exactly per
    jump_consistent_hash(cdbhash(o_custkey)); no qualifying row lost."""
    sf = 0.05
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    o = orc.gen_orders(sf)
    for nsegs in (2, 8):
        counts, rows = ctx.test_motion1(ordr, nsegs)
        qual = o["o_orderdate"] < orc.CUTOFF_19950315
        assert len(rows) == int(qual.sum())
        assert counts.sum() == len(rows)
        # per-destination regions contain exactly the rows routing there
        want_dest = orc.route(o["o_custkey"][qual], nsegs)
        got_dest = orc.route(rows["ocust"], nsegs)
        off = 0
        for d in range(nsegs):
            seg_rows = rows[off:off + counts[d]]
            assert (orc.route(seg_rows["ocust"], nsegs) == d).all()
            off += counts[d]
        # same multiset of rows overall
        assert (np.sort(rows["okey"]) == np.sort(o["o_orderkey"][qual])).all()
        np.testing.assert_array_equal(np.bincount(got_dest, minlength=nsegs),
                                      np.bincount(want_dest, minlength=nsegs))
    ordr.free()


# ---------------- edge cases ----------------

def test_q3_empty_result(ctx, orc):
    """Cutoff before every orderdate → zero qualifying orders, zero groups."""
    sf = 0.01
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    early = orc.lib.orc_date_adt(1990, 1, 1)
    q = ctx.q3(cust, ordr, li, cutoff=early).run()
    got = q.result()
    assert len(got["l_orderkey"]) == 0
    q.free()
    # and a cutoff after every shipdate → lineitem filter kills everything
    late = orc.lib.orc_date_adt(2001, 1, 1)
    q2 = ctx.q3(cust, ordr, li, cutoff=late).run()
    # every order qualifies but no lineitem passes shipdate > cutoff
    assert len(q2.result()["l_orderkey"]) == 0
    q2.free(); li.free(); ordr.free(); cust.free()


def test_q3_rerun_deterministic(ctx):
    sf = 0.05
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    q = ctx.q3(cust, ordr, li)
    a = q.run().result()
    b = q.run().result()
    np.testing.assert_array_equal(a["l_orderkey"], b["l_orderkey"])
    np.testing.assert_array_equal(a["nitems"], b["nitems"])
    # identical group sets; f64 sums may differ in last bits across runs
    np.testing.assert_allclose(a["revenue"], b["revenue"], rtol=1e-12)
    q.free(); li.free(); ordr.free(); cust.free()


# ---------------- size-independent properties at large size ----------------

def test_q3_properties_sf10(ctx, orc):
    """At sizes the oracle can't cover in seconds, check invariants:
    linearity of the revenue sum vs a per-group recomputation on sampled
    groups, total probe hits vs the count sum, and determinism."""
    sf = 10.0
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    q = ctx.q3(cust, ordr, li).run()
    got = q.result()
    st = q.stats()
    assert st["probe_hits"] == int(got["nitems"].sum())
    assert st["groups"] == len(got["l_orderkey"])
    assert len(np.unique(got["l_orderkey"])) == len(got["l_orderkey"])
    # spot-check 64 groups against direct recomputation from the generators
    rng = np.random.default_rng(1)
    idx = rng.choice(len(got["l_orderkey"]), 64, replace=False)
    for i in idx:
        o = int(got["l_orderkey"][i])
        nl = orc.lib.orc_mix(42, 5, o) % 7 + 1
        rev = 0.0
        cnt = 0
        for j in range(nl):
            ship = -2921 + orc.lib.orc_mix(42, 6, o * 8 + j) % 2526
            if ship > gx.CUTOFF_19950315:
                price = (90000 + orc.lib.orc_mix(42, 7, o * 8 + j) % 10410001) / 100.0
                disc = (orc.lib.orc_mix(42, 8, o * 8 + j) % 11) / 100.0
                rev += price * (1.0 - disc)
                cnt += 1
        assert cnt == got["nitems"][i]
        np.testing.assert_allclose(rev, got["revenue"][i], rtol=1e-9)
    q.free(); li.free(); ordr.free(); cust.free()


# ---------------- numeric(15,2) mode (bit-exact aggregation) ----------------

def test_q3_numeric_bit_exact(ctx, orc):
    """With scaled-int64 measures the ENTIRE result is bit-exact vs the
    oracle — including the aggregate (integer atomics are order-free)."""
    sf = 0.2
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM_NUMERIC, sf)
    q = ctx.q3(cust, ordr, li, numeric=True).run()
    got = q.result()
    want = orc.q3_numeric(orc.gen_customer(sf), orc.gen_orders(sf),
                          orc.gen_lineitem(sf))
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["o_orderdate"], want["o_orderdate"])
    np.testing.assert_array_equal(got["revenue_num"], want["revenue_num"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    # and a re-run is IDENTICAL bit-for-bit (integer aggregation)
    again = q.run().result()
    np.testing.assert_array_equal(got["revenue_num"], again["revenue_num"])
    q.free(); li.free(); ordr.free(); cust.free()


# ---------------- fused Q3 scan over RLE-compressed l_orderkey ----------------

def test_q3_rle_keys_parity(ctx, orc):
    """The run-iterating probe kernel over an RLE l_orderkey column must give
    identical results to the flat-scan pipeline and the oracle."""
    for sf in (0.01, 0.3):
        cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
        ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
        li = ctx.tpch_gen(gx.TPCH_LINEITEM_RLEKEY, sf)
        # the RLE stream itself decodes to the oracle's keys
        want_li = orc.gen_lineitem(sf)
        np.testing.assert_array_equal(li.decode_column(0, np.int64, verify=True),
                                      want_li["l_orderkey"])
        q = ctx.q3(cust, ordr, li).run()
        got = q.result()
        want = orc.q3(orc.gen_customer(sf), orc.gen_orders(sf), want_li)
        np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
        np.testing.assert_array_equal(got["nitems"], want["nitems"])
        np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-6)
        assert q.stats()["probe_hits"] == int(want["nitems"].sum())
        q.free(); li.free(); ordr.free(); cust.free()


def test_delta_stream_gpu_decode(ctx, orc):
    """RLE+DELTA Dense_Enhanced streams decode on device (serial per-block
    walker) to the original values."""
    rng = np.random.default_rng(21)
    keys = np.repeat(np.arange(1, 80000, dtype=np.int64),
                     rng.integers(1, 8, 79999))
    n = len(keys)
    zig = np.cumsum(rng.integers(-50, 50, n)).astype(np.int32)
    t = ctx.bind([(orc.aocs_encode_rle_delta(keys), 8, n, 1),
                  (orc.aocs_encode_rle_delta(zig), 4, n, 1)])
    np.testing.assert_array_equal(t.decode_column(0, np.int64, verify=True), keys)
    np.testing.assert_array_equal(t.decode_column(1, np.int32, verify=True), zig)
    t.free()


def test_zlib_stream_bind_and_full_q3(ctx, orc):
    """zlib bulk-compressed streams bind transparently (host inflate at bind,
    like the reference's CPU-side decompress-on-read) — and a FULL Q3 runs on
    tables bound from oracle-encoded compressed streams (the extension-mode
    ingest path) with exact parity."""
    sf = 0.05
    c = orc.gen_customer(sf)
    o = orc.gen_orders(sf)
    li = orc.gen_lineitem(sf)
    tc = ctx.bind([(orc.aocs_encode_zlib(c["c_custkey"]), 8, len(c["c_custkey"])),
                   (orc.aocs_encode(c["c_mktsegment"].view(np.int8)), 1, len(c["c_custkey"]))])
    to = ctx.bind([(orc.aocs_encode_zlib(o["o_orderkey"]), 8, len(o["o_orderkey"])),
                   (orc.aocs_encode_zlib(o["o_custkey"]), 8, len(o["o_orderkey"])),
                   (orc.aocs_encode_zlib(o["o_orderdate"]), 4, len(o["o_orderkey"])),
                   (orc.aocs_encode(o["o_shippriority"]), 4, len(o["o_orderkey"]))])
    tl = ctx.bind([(orc.aocs_encode_zlib(li["l_orderkey"]), 8, len(li["l_orderkey"])),
                   (orc.aocs_encode(li["l_extendedprice"]), 8, len(li["l_orderkey"])),
                   (orc.aocs_encode(li["l_discount"]), 8, len(li["l_orderkey"])),
                   (orc.aocs_encode_zlib(li["l_shipdate"]), 4, len(li["l_orderkey"]))])
    # decode check on one compressed column
    np.testing.assert_array_equal(to.decode_column(2, np.int32, verify=True),
                                  o["o_orderdate"])
    q = ctx.q3(tc, to, tl).run()
    got = q.result()
    want = orc.q3(c, o, li)
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-6)
    q.free(); tl.free(); to.free(); tc.free()


def test_motion_full_pipeline_one_gpu(ctx, orc):
    """The COMPLETE nsegs>1 pipeline on one GPU, with the RCCL exchange
    replaced by host routing of the packed rows (the exchange itself moves
    bytes verbatim): per-rank Motion-1 partition → Motion-2 semijoin+route →
    table build from received rows → probe each rank's lineitem shard.
    Union of per-rank groups must equal the global single-segment result."""
    nsegs, sf = 3, 0.05
    glob = orc.q3(orc.gen_customer(sf), orc.gen_orders(sf), orc.gen_lineitem(sf))

    # tables are sharded by generating with seg/nsegs through a throw-away
    # context; the kernels themselves are seg-agnostic
    m1_out = {r: [] for r in range(nsegs)}      # rows received by rank r
    for r in range(nsegs):
        # rank r's orders shard lives where o_orderkey routes; partition its
        # filtered rows by o_custkey (Motion 1)
        cshard = gx.Context(device=0, seg=r, nsegs=nsegs)
        ordr = cshard.tpch_gen(gx.TPCH_ORDERS, sf)
        counts, rows = cshard.test_motion1(ordr, nsegs)
        off = 0
        for d in range(nsegs):
            m1_out[d].append(rows[off:off + counts[d]])
            off += counts[d]
        ordr.free(); cshard.close()
    qual_by_dest = {r: [] for r in range(nsegs)}
    for r in range(nsegs):
        cshard = gx.Context(device=0, seg=r, nsegs=nsegs)
        cust = cshard.tpch_gen(gx.TPCH_CUSTOMER, sf)
        recv = np.concatenate(m1_out[r]) if m1_out[r] else np.zeros(0)
        counts, qual = cshard.test_qual(cust, recv, nsegs)
        off = 0
        for d in range(nsegs):
            qual_by_dest[d].append(qual[off:off + counts[d]])
            off += counts[d]
        cust.free(); cshard.close()
    results = []
    for r in range(nsegs):
        cshard = gx.Context(device=0, seg=r, nsegs=nsegs)
        li = cshard.tpch_gen(gx.TPCH_LINEITEM, sf)
        qual = np.concatenate(qual_by_dest[r])
        results.append(cshard.test_q3_from_qual(qual, li))
        li.free(); cshard.close()
    keys = np.concatenate([x["l_orderkey"] for x in results])
    rev = np.concatenate([x["revenue"] for x in results])
    cnt = np.concatenate([x["nitems"] for x in results])
    order = np.argsort(keys)
    assert len(keys) == len(glob["l_orderkey"])
    np.testing.assert_array_equal(keys[order], glob["l_orderkey"])
    np.testing.assert_array_equal(cnt[order], glob["nitems"])
    np.testing.assert_allclose(rev[order], glob["revenue"], rtol=1e-6)


def test_zstd_stream_bind(ctx, orc):
    """zstd bulk-compressed streams bind transparently (codec=2)."""
    vals = np.repeat(np.arange(1, 30000, dtype=np.int64), 3)
    t = ctx.bind([(orc.aocs_encode_zstd(vals), 8, len(vals), 0, 2)])
    np.testing.assert_array_equal(t.decode_column(0, np.int64, verify=True), vals)
    t.free()


def test_scan_filter_counts(ctx, orc):
    """Standalone SeqScan+qual (BASELINE config 2 semantics): counts on every
    op match numpy over the oracle's columns."""
    sf = 0.2
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    want = orc.gen_lineitem(sf)
    cut = orc.CUTOFF_19950315
    n, _ = li.scan_filter(3, ">", cut)            # l_shipdate > cutoff
    assert n == int((want["l_shipdate"] > cut).sum())
    n, _ = li.scan_filter(3, "<", cut)
    assert n == int((want["l_shipdate"] < cut).sum())
    n, _ = li.scan_filter(0, "==", 12345)         # l_orderkey = const
    assert n == int((want["l_orderkey"] == 12345).sum())
    n, _ = li.scan_filter(0, "!=", 12345)
    assert n == int((want["l_orderkey"] != 12345).sum())
    li.free()
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    wc = orc.gen_customer(sf)
    n, _ = cust.scan_filter(1, "==", 0)           # c_mktsegment = BUILDING
    assert n == int((wc["c_mktsegment"] == 0).sum())
    cust.free()


def test_q1_core_parity(ctx, orc):
    """BASELINE config 4: GROUP BY l_returnflag,l_linestatus SUM/AVG/COUNT —
    counts bit-exact, f64 sums within 1e-6 (config tolerance)."""
    sf = 0.3
    cut = orc.lib.orc_date_adt(1998, 9, 2)   # Q1's shipdate <= date cutoff
    li = ctx.tpch_gen(gx.TPCH_LINEITEM_Q1, sf)
    got = ctx.q1(li, cut)
    want = orc.q1(sf, cut)
    np.testing.assert_array_equal(got["count"], want["count"])
    np.testing.assert_allclose(got["sum_price"], want["sum_price"], rtol=1e-6)
    np.testing.assert_allclose(got["sum_revenue"], want["sum_revenue"], rtol=1e-6)
    assert got["count"].sum() > 1000
    li.free()


def test_rccl_collective_selfcheck(ctx):
    """RCCL init + one collective on this box (1-rank communicator): the
    multi-rank exchange differs only in peer count."""
    import ctypes
    lib = gx.lib()
    lib.gx_selftest_rccl.restype = ctypes.c_int
    lib.gx_selftest_rccl.argtypes = [ctypes.c_int]
    assert lib.gx_selftest_rccl(0) == 0


def test_q3_descriptor_api(ctx, orc):
    """The plan-descriptor entry (gx_q3_prepare_desc) with the standard Q3
    roles equals the classic entry; with FLIPPED filter ops it matches a
    numpy brute force of the altered plan — the descriptor really drives
    the kernels."""
    sf = 0.05
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    cut = gx.CUTOFF_19950315
    base = {"dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", cut),
            "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, ">", cut)}
    classic = ctx.q3(cust, ordr, li).run().result()
    via_desc = ctx.q3_desc(base).run().result()
    np.testing.assert_array_equal(via_desc["l_orderkey"], classic["l_orderkey"])
    np.testing.assert_array_equal(via_desc["nitems"], classic["nitems"])
    # f64 atomic ordering varies run-to-run (documented tolerance)
    np.testing.assert_allclose(via_desc["revenue"], classic["revenue"], rtol=1e-12)

    # altered plan: segment != 0, orderdate >= cut, shipdate <= cut
    alt = dict(base)
    alt["dim_filter"] = (1, "!=", 0)
    alt["mid_filter"] = (2, ">=", cut)
    alt["fact_filter"] = (3, "<=", cut)
    got = ctx.q3_desc(alt).run().result()
    c = orc.gen_customer(sf)
    o = orc.gen_orders(sf)
    w = orc.gen_lineitem(sf)
    segok = c["c_custkey"][c["c_mktsegment"] != 0]
    om = (o["o_orderdate"] >= cut) & np.isin(o["o_custkey"], segok)
    okeys = set(o["o_orderkey"][om].tolist())
    lm = (w["l_shipdate"] <= cut) & np.isin(w["l_orderkey"],
                                            o["o_orderkey"][om])
    keys, counts = np.unique(w["l_orderkey"][lm], return_counts=True)
    np.testing.assert_array_equal(got["l_orderkey"], keys)
    np.testing.assert_array_equal(got["nitems"], counts)
    rev = {k: 0.0 for k in keys.tolist()}
    for k, p, dsc in zip(w["l_orderkey"][lm].tolist(),
                         w["l_extendedprice"][lm], w["l_discount"][lm]):
        rev[k] += p * (1.0 - dsc)
    np.testing.assert_allclose(got["revenue"],
                               np.array([rev[k] for k in keys.tolist()]),
                               rtol=1e-6)
    li.free(); ordr.free(); cust.free()


def test_gpu_decodes_reference_writer_blocks(ctx, orc):
    """The GPU decoder over blocks written by the REFERENCE's own
    datumstreamblock.c (rle_type + delta_range)."""
    import os
    here = os.path.dirname(os.path.abspath(__file__))
    rng = np.random.default_rng(20260915)
    keys = np.repeat(np.arange(1, 12000, dtype=np.int64),
                     rng.integers(1, 8, 11999))
    s = open(os.path.join(here, "golden", "refwriter_rle_delta_i64.bin"), "rb").read()
    t = ctx.bind([(s, 8, len(keys), 1)])
    np.testing.assert_array_equal(t.decode_column(0, np.int64, verify=True), keys)
    t.free()


def test_gpu_encoder_byte_exact_vs_oracle(ctx, orc):
    """The GPU encoder's device streams byte-equal the oracle encoder's
    (which is itself byte-exact with the reference writer) — so the judged
    bench scans streams a real Cloudberry segment would write."""
    sf = 0.05
    t = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    o = orc.gen_orders(sf)
    for col, arr in [(0, o["o_orderkey"]), (1, o["o_custkey"]),
                     (2, o["o_orderdate"]), (3, o["o_shippriority"])]:
        assert t.dump_stream(col) == orc.aocs_encode(arr), f"col {col}"
    t.free()


@pytest.mark.gpu
def test_gpu_nullable_decode_parity(ctx, orc):
    """GPU decode of NULL-bearing blocks (Orig and Dense RLE+DELTA) matches
    the oracle's nullable decoder bit-for-bit on values AND validity; the
    oracle streams are themselves byte-exact with the reference writer
    (test_oracle_cpu.py)."""
    rng = np.random.default_rng(23)
    vals = np.repeat(np.arange(1, 5000, dtype=np.int64),
                     rng.integers(1, 9, 4999))[:30000]
    nulls = (rng.random(len(vals)) < 0.15).astype(np.uint8)
    v32 = rng.integers(-3000, 3000, 40000).astype(np.int32)
    n32 = (rng.random(40000) < 0.5).astype(np.uint8)
    cases = [
        (orc.aocs_encode_rle_delta_nulls(vals, nulls), 8, vals, nulls, np.int64),
        (orc.aocs_encode_orig_nulls(vals, nulls), 8, vals, nulls, np.int64),
        (orc.aocs_encode_rle_delta_nulls(v32, n32), 4, v32, n32, np.int32),
    ]
    for stream, width, v, nl, dt in cases:
        t = ctx.bind([(stream, width, len(v), 1)])
        out, valid = t.decode_column_nullable(0, dt, verify=True)
        np.testing.assert_array_equal(valid, (nl == 0).astype(np.uint8))
        m = nl == 0
        np.testing.assert_array_equal(out[m], v[m])
        assert (out[~m] == 0).all()
        # oracle cross-check on the same bytes
        oo, ov = orc.aocs_decode_nullable(stream, width, len(v), dt)
        np.testing.assert_array_equal(out, oo)
        np.testing.assert_array_equal(valid, ov)
        t.free()


@pytest.mark.gpu
def test_gpu_plain_decode_refuses_null_blocks(ctx, orc):
    """gx_decode_column (no validity) must fail loudly on a NULL-bearing
    block, never silently return garbage."""
    vals = np.arange(2000, dtype=np.int64)
    nulls = np.zeros(2000, np.uint8)
    nulls[7] = 1
    t = ctx.bind([(orc.aocs_encode_rle_delta_nulls(vals, nulls), 8, 2000, 1)])
    with pytest.raises(gx.GxError):
        t.decode_column(0, np.int64)
    t.free()


@pytest.mark.gpu
def test_gpu_forced_motion_path_matches_local(ctx, orc):
    """GX_FORCE_MOTION=1 routes a 1-segment run through the FULL RCCL
    exchange branch (count AllGather + grouped self send/recv + table
    rebuild from received rows) — the exact code the 8-GPU scale bench
    executes.  Results must equal the local-path run and the oracle."""
    import os
    sf = 0.5
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    want = ctx.q3(cust, ordr, li).run().result()

    ctx.comm_init(ctx.comm_unique_id())
    os.environ["GX_FORCE_MOTION"] = "1"
    try:
        got = ctx.q3(cust, ordr, li).run().result()
    finally:
        del os.environ["GX_FORCE_MOTION"]
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["o_orderdate"], want["o_orderdate"])
    np.testing.assert_array_equal(got["o_shippriority"], want["o_shippriority"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-12)


@pytest.mark.gpu
def test_gpu_visimap_q3_parity(ctx, orc):
    """AO visimap executor semantics (cdbappendonlyvisimap.c): scans skip
    hidden tuples on every table.  GPU Q3 with random deletes on customer,
    orders AND lineitem must equal the oracle run on the kept rows only."""
    sf = 0.2
    rng = np.random.default_rng(31)
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    dc = rng.random(cust.nrows) < 0.1
    do = rng.random(ordr.nrows) < 0.2
    dl = rng.random(li.nrows) < 0.15
    cust.set_visimap(dc)
    ordr.set_visimap(do)
    li.set_visimap(dl)
    got = ctx.q3(cust, ordr, li).run().result()

    c = orc.gen_customer(sf)
    o = orc.gen_orders(sf)
    l = orc.gen_lineitem(sf)
    c = {k: v[~dc] for k, v in c.items()}
    o = {k: v[~do] for k, v in o.items()}
    l = {k: v[~dl] for k, v in l.items()}
    want = orc.q3(c, o, l)
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["o_orderdate"], want["o_orderdate"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-9)

    # clearing restores the full result
    cust.set_visimap(None)
    ordr.set_visimap(None)
    li.set_visimap(None)
    full = ctx.q3(cust, ordr, li).run().result()
    want_full = orc.q3(orc.gen_customer(sf), orc.gen_orders(sf),
                       orc.gen_lineitem(sf))
    np.testing.assert_array_equal(full["l_orderkey"], want_full["l_orderkey"])


@pytest.mark.gpu
def test_gpu_visimap_rle_and_scan(ctx, orc):
    """Visimap on the RLE fused-scan path and on gx_scan_filter."""
    sf = 0.1
    rng = np.random.default_rng(37)
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM_RLEKEY, sf)
    dl = rng.random(li.nrows) < 0.25
    li.set_visimap(dl)
    got = ctx.q3(cust, ordr, li).run().result()
    c = orc.gen_customer(sf)
    o = orc.gen_orders(sf)
    l = orc.gen_lineitem(sf)
    l = {k: v[~dl] for k, v in l.items()}
    want = orc.q3(c, o, l)
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-9)

    # scan_filter skips hidden rows
    t = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    dd = rng.random(t.nrows) < 0.5
    odate = orc.gen_orders(sf)["o_orderdate"]
    n_all, _ = t.scan_filter(2, "<", gx.CUTOFF_19950315)
    t.set_visimap(dd)
    n_vis, _ = t.scan_filter(2, "<", gx.CUTOFF_19950315)
    assert n_all == int((odate < gx.CUTOFF_19950315).sum())
    assert n_vis == int((odate[~dd] < gx.CUTOFF_19950315).sum())


@pytest.mark.gpu
def test_gpu_multi_segfile_concat_decode(ctx, orc):
    """GPU dir-based decode of a column concatenated from three segment
    files' streams (aocsam.c open_next_scan_seg semantics)."""
    rng = np.random.default_rng(43)
    vals = np.repeat(np.arange(1, 7000, dtype=np.int64),
                     rng.integers(1, 9, 6999))
    cut = len(vals) // 3
    s = (orc.aocs_encode_rle_delta(vals[:cut])
         + orc.aocs_encode_rle_delta(vals[cut:2 * cut])
         + orc.aocs_encode_rle_delta(vals[2 * cut:]))
    t = ctx.bind([(s, 8, len(vals), 1)])
    np.testing.assert_array_equal(t.decode_column(0, np.int64, verify=True),
                                  vals)
    t.free()


@pytest.mark.gpu
def test_gpu_empty_tables(ctx, orc):
    """Empty tables (zero AO blocks) through gen, bind, decode and the
    full Q3 pipeline — the uao_* regress edge the reference tests."""
    c = ctx.tpch_gen(gx.TPCH_CUSTOMER, 0.0)
    o = ctx.tpch_gen(gx.TPCH_ORDERS, 0.0)
    l = ctx.tpch_gen(gx.TPCH_LINEITEM, 0.0)
    assert (c.nrows, o.nrows, l.nrows) == (0, 0, 0)
    assert len(ctx.q3(c, o, l).run().result()["l_orderkey"]) == 0
    # one empty side only
    c2 = ctx.tpch_gen(gx.TPCH_CUSTOMER, 0.05)
    o2 = ctx.tpch_gen(gx.TPCH_ORDERS, 0.05)
    assert len(ctx.q3(c2, o2, l).run().result()["l_orderkey"]) == 0
    # empty bind + decode, both formats
    for fmt in (0, 1):
        t = ctx.bind([(b"", 8, 0, fmt)])
        assert t.nrows == 0
        assert t.decode_column(0, np.int64).shape == (0,)
        t.free()
    # single-row stream end-to-end
    s = orc.aocs_encode_rle_delta(np.array([7], np.int64))
    t = ctx.bind([(s, 8, 1, 1)])
    np.testing.assert_array_equal(t.decode_column(0, np.int64, verify=True),
                                  [7])
    t.free()


@pytest.mark.gpu
def test_gpu_varlena_decode_parity(ctx, orc):
    """GPU varlena decode (two-pass per-block walk) matches the oracle on
    reference-writer-exact streams, with and without NULLs."""
    rng = np.random.default_rng(59)
    words = [b"BUILDING", b"AUTOMOBILE", b"MACHINERY"]
    strings = [words[i % 3] + b"#" + str(i * 7).encode() for i in range(30000)]
    for i in range(0, 30000, 41):
        strings[i] = bytes(rng.integers(97, 122,
                                        int(rng.integers(127, 300)))
                           .astype(np.uint8))
    s = orc.aocs_encode_varlena(strings)
    t = ctx.bind([(s, -1, len(strings), 1)])
    assert t.decode_column_varlena(0, verify=True) == strings
    t.free()
    nulls = (rng.random(30000) < 0.2).astype(np.uint8)
    s2 = orc.aocs_encode_varlena(strings, nulls)
    t2 = ctx.bind([(s2, -1, len(strings), 1)])
    got = t2.decode_column_varlena(0, verify=True)
    assert got == [None if nulls[i] else strings[i] for i in range(30000)]
    t2.free()
    # Dense rle_type varlena (incl. RLE expansion past the stream size)
    reps = []
    for i, r in enumerate(rng.integers(1, 60, 1500)):
        reps += [[b"BUILDING", b"MACHINERY", b"HOUSEHOLD"][i % 3]] * int(r)
    s3 = orc.aocs_encode_varlena_rle(reps, nulls=(rng.random(len(reps)) < 0.1
                                                  ).astype(np.uint8))
    want3 = orc.aocs_decode_varlena(s3, len(reps))
    t3 = ctx.bind([(s3, -1, len(reps), 1)])
    assert t3.decode_column_varlena(0, verify=True) == want3
    t3.free()


@pytest.mark.gpu
def test_gpu_text_dim_predicate_q3(ctx, orc):
    """The reference's ACTUAL Q3 dim qual — c_mktsegment = 'BUILDING' as a
    TEXT predicate (texteq) evaluated on a varlena rle_type column inside
    the pipeline (one comparison per run) — matches the oracle run on the
    equivalent code-filtered tables."""
    sf = 0.5
    segs = [b"BUILDING", b"AUTOMOBILE", b"MACHINERY", b"HOUSEHOLD",
            b"FURNITURE"]
    c = orc.gen_customer(sf)
    strings = [segs[int(x)] for x in c["c_mktsegment"]]
    cust = ctx.bind([
        (orc.aocs_encode(c["c_custkey"]), 8, len(strings), 0),
        (orc.aocs_encode_varlena_rle(strings), -1, len(strings), 1),
    ])
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    q = ctx.q3_desc({
        "dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", "BUILDING"),
        "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
        "mid_attr1_col": 2, "mid_attr2_col": 3,
        "mid_filter": (2, "<", float(gx.CUTOFF_19950315)),
        "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
        "fact_filter": (3, ">", float(gx.CUTOFF_19950315)),
    }).run()
    got = q.result()
    want = orc.q3(c, orc.gen_orders(sf), orc.gen_lineitem(sf))
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["o_orderdate"], want["o_orderdate"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-9)
    # a non-matching literal selects nothing
    q2 = ctx.q3_desc({
        "dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", "NOSEGMENT"),
        "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
        "mid_attr1_col": 2, "mid_attr2_col": 3,
        "mid_filter": (2, "<", float(gx.CUTOFF_19950315)),
        "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
        "fact_filter": (3, ">", float(gx.CUTOFF_19950315)),
    }).run()
    assert len(q2.result()["l_orderkey"]) == 0


@pytest.mark.gpu
def test_gpu_visimap_q1(ctx, orc):
    """Visimap on the Q1 aggregation path: hidden lineitems are excluded
    from every group aggregate."""
    sf = 0.1
    rng = np.random.default_rng(67)
    t = ctx.tpch_gen(gx.TPCH_LINEITEM_Q1, sf)
    cutoff = -200
    base = ctx.q1(t, cutoff)
    dl = rng.random(t.nrows) < 0.3
    t.set_visimap(dl)
    vis = ctx.q1(t, cutoff)
    assert vis["count"].sum() < base["count"].sum()
    # oracle check: rebuild the expected groups from generated rows
    li = orc.gen_lineitem_q1(sf) if hasattr(orc, "gen_lineitem_q1") else None
    if li is None:
        # derive expected by re-running with the complement: visible+hidden
        t.set_visimap(None)
        again = ctx.q1(t, cutoff)
        np.testing.assert_array_equal(again["count"], base["count"])
        return
    m = (~dl) & (li["l_shipdate"] <= cutoff)
    g = li["l_returnflag"].astype(np.int64) * 2 + li["l_linestatus"]
    for gi in range(6):
        sel = m & (g == gi)
        assert int(sel.sum()) == int(vis["count"][gi])


@pytest.mark.gpu
def test_gpu_text_desc_validation(ctx, orc):
    """TEXT dim predicates are validated loudly: wrong column type or a
    non-equality op is refused, not silently mis-planned."""
    sf = 0.05
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)   # mkt col is int8, NOT text
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    with pytest.raises(gx.GxError):
        ctx.q3_desc({
            "dim": cust, "dim_key_col": 0,
            "dim_filter": (1, "==", "BUILDING"),   # text literal on int8 col
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3,
            "mid_filter": (2, "<", gx.CUTOFF_19950315),
            "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, ">", gx.CUTOFF_19950315),
        })


@pytest.mark.gpu
def test_gpu_u64_keys_q3(ctx, orc):
    """Keys above 2^32 force the u64 cset/table instantiations and (sparse
    key range) the HASH slot mapping instead of interpolation — the paths
    small TPC-H keys never reach."""
    sf = 0.1
    SHIFT_C = 1 << 33
    SHIFT_O = 1 << 34
    c = orc.gen_customer(sf)
    o = orc.gen_orders(sf)
    l = orc.gen_lineitem(sf)
    c2 = {"c_custkey": c["c_custkey"] + SHIFT_C,
          "c_mktsegment": c["c_mktsegment"]}
    o2 = {"o_orderkey": o["o_orderkey"] + SHIFT_O,
          "o_custkey": o["o_custkey"] + SHIFT_C,
          "o_orderdate": o["o_orderdate"],
          "o_shippriority": o["o_shippriority"]}
    l2 = {"l_orderkey": l["l_orderkey"] + SHIFT_O,
          "l_extendedprice": l["l_extendedprice"],
          "l_discount": l["l_discount"], "l_shipdate": l["l_shipdate"]}
    cust = ctx.bind([(orc.aocs_encode(c2["c_custkey"]), 8, len(c["c_custkey"]), 0),
                     (orc.aocs_encode(c2["c_mktsegment"].astype(np.int8)), 1,
                      len(c["c_custkey"]), 0)])
    ordr = ctx.bind([(orc.aocs_encode(o2["o_orderkey"]), 8, len(o["o_orderkey"]), 0),
                     (orc.aocs_encode(o2["o_custkey"]), 8, len(o["o_orderkey"]), 0),
                     (orc.aocs_encode(o2["o_orderdate"]), 4, len(o["o_orderkey"]), 0),
                     (orc.aocs_encode(o2["o_shippriority"]), 4, len(o["o_orderkey"]), 0)])
    li = ctx.bind([(orc.aocs_encode(l2["l_orderkey"]), 8, len(l["l_orderkey"]), 0),
                   (orc.aocs_encode(l2["l_extendedprice"].view(np.int64)), 8,
                    len(l["l_orderkey"]), 0),
                   (orc.aocs_encode(l2["l_discount"].view(np.int64)), 8,
                    len(l["l_orderkey"]), 0),
                   (orc.aocs_encode(l2["l_shipdate"]), 4, len(l["l_orderkey"]), 0)])
    got = ctx.q3(cust, ordr, li).run().result()
    want = orc.q3(c2, o2, l2)
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-9)

    # same big-key plan through the full RCCL exchange branch (u64 motion
    # table rebuild); comm may already exist from an earlier test
    import os
    try:
        ctx.comm_init(ctx.comm_unique_id())
    except gx.GxError:
        pass
    os.environ["GX_FORCE_MOTION"] = "1"
    try:
        got2 = ctx.q3(cust, ordr, li).run().result()
    finally:
        del os.environ["GX_FORCE_MOTION"]
    np.testing.assert_array_equal(got2["l_orderkey"], want["l_orderkey"])
    np.testing.assert_allclose(got2["revenue"], want["revenue"], rtol=1e-9)


@pytest.mark.gpu
def test_gpu_sf100_full_size_properties(ctx, orc):
    """Size-independent properties at the FULL judged size (SF100, 600M
    lineitem rows — SURVEY §8c): group keys strictly ascending, per-group
    counts consistent with the probe's hit counter, and the f64 SUM
    cross-validated against the INDEPENDENT scaled-int64 (numeric mode)
    aggregation — two arithmetic paths over the same 600M rows must agree
    to f64 rounding."""
    sf = 100.0
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    q = ctx.q3(cust, ordr, li).run()
    r = q.result()
    st = q.stats()
    keys = r["l_orderkey"]
    assert (np.diff(keys) > 0).all()                  # sorted, unique
    assert (r["nitems"] >= 1).all()
    assert int(r["nitems"].sum()) == int(st["probe_hits"])  # conserve hits
    assert (r["revenue"] > 0).all()

    li_num = ctx.tpch_gen(gx.TPCH_LINEITEM_NUMERIC, sf)
    qn = ctx.q3(cust, ordr, li_num, numeric=True).run()
    rn = qn.result()
    np.testing.assert_array_equal(rn["l_orderkey"], keys)
    np.testing.assert_array_equal(rn["nitems"], r["nitems"])
    # numeric revenue is an EXACT integer with implied scale 1e-4
    np.testing.assert_allclose(rn["revenue_num"] * 1e-4, r["revenue"],
                               rtol=1e-9)


@pytest.mark.gpu
def test_gpu_sf100_rle_mode_cross_check(ctx, orc):
    """Full-size (SF100) cross-validation of the FUSED RLE probe: Q3 over
    rle_type l_orderkey must equal Q3 over the plain stream — same 600M
    rows through two different scan kernels."""
    sf = 100.0
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    r = ctx.q3(cust, ordr, li).run().result()
    li_rle = ctx.tpch_gen(gx.TPCH_LINEITEM_RLEKEY, sf)
    rr = ctx.q3(cust, ordr, li_rle).run().result()
    np.testing.assert_array_equal(rr["l_orderkey"], r["l_orderkey"])
    np.testing.assert_array_equal(rr["nitems"], r["nitems"])
    np.testing.assert_allclose(rr["revenue"], r["revenue"], rtol=1e-12)


# ---------------- round-2 guards (ADVICE r01 / VERDICT r01 #7) ----------------

def _mini_q3_tables(ctx, orc, li_keys, li_price, li_disc, li_ship,
                    c_keys=None, o_keys=None, o_cust=None):
    """Hand-built Q3-shaped tables through gx_table_bind: every customer is
    segment 0 (passes == 0), every order qualifies on date."""
    if c_keys is None:
        c_keys = np.arange(1, 101, dtype=np.int64)
    if o_keys is None:
        o_keys = np.arange(1, 101, dtype=np.int64)
    if o_cust is None:
        o_cust = c_keys[:len(o_keys)].copy()
    nc, no = len(c_keys), len(o_keys)
    cust = ctx.bind([(orc.aocs_encode(c_keys), 8, nc),
                     (orc.aocs_encode(np.zeros(nc, np.int8)), 1, nc)])
    ordr = ctx.bind([(orc.aocs_encode(o_keys), 8, no),
                     (orc.aocs_encode(o_cust), 8, no),
                     (orc.aocs_encode(np.full(no, -9999, np.int32)), 4, no),
                     (orc.aocs_encode(np.arange(no, dtype=np.int32)), 4, no)])
    nl = len(li_keys)
    li = ctx.bind([(orc.aocs_encode(li_keys), 8, nl),
                   (orc.aocs_encode(li_price), 8, nl),
                   (orc.aocs_encode(li_disc), 8, nl),
                   (orc.aocs_encode(li_ship), 4, nl)])
    return cust, ordr, li


def test_u32_narrow_mode_straddling_keys(ctx, orc):
    """with all BUILD keys < 2^32 the table/set use u32
    slots; PROBE keys >= 2^32 whose low half collides with a resident key
    must MISS, not aggregate into the wrong group."""
    small = np.arange(1, 101, dtype=np.int64)
    straddle = small + (1 << 32)          # low 32 bits identical to `small`
    li_keys = np.concatenate([small, straddle])
    nl = len(li_keys)
    price = np.full(nl, 100.0)
    disc = np.zeros(nl)
    ship = np.full(nl, 9999, np.int32)    # all pass '>' cutoff
    cust, ordr, li = _mini_q3_tables(ctx, orc, li_keys, price, disc, ship)
    r = ctx.q3(cust, ordr, li).run().result()
    # only the 100 small keys may appear, each with exactly ONE item
    np.testing.assert_array_equal(r["l_orderkey"], small)
    np.testing.assert_array_equal(r["nitems"], np.ones(100, np.int64))
    np.testing.assert_allclose(r["revenue"], np.full(100, 100.0), rtol=1e-12)
    li.free(); ordr.free(); cust.free()

    # same hazard on the customer SET probe: o_custkey = c_custkey + 2^32
    # must not pass the semijoin
    c_keys = np.arange(1, 101, dtype=np.int64)
    o_keys = np.arange(1, 201, dtype=np.int64)
    o_cust = np.concatenate([c_keys, c_keys + (1 << 32)])
    li_keys2 = np.arange(1, 201, dtype=np.int64)
    cust, ordr, li = _mini_q3_tables(
        ctx, orc, li_keys2, np.full(200, 10.0), np.zeros(200),
        np.full(200, 9999, np.int32), c_keys=c_keys, o_keys=o_keys,
        o_cust=o_cust)
    r = ctx.q3(cust, ordr, li).run().result()
    # orders 101..200 carry straddling custkeys -> filtered by the semijoin
    np.testing.assert_array_equal(r["l_orderkey"],
                                  np.arange(1, 101, dtype=np.int64))
    li.free(); ordr.free(); cust.free()


def test_key_zero_rejected(ctx, orc):
    """Join key 0 collides with the empty-slot sentinel: sizing must reject
    it loudly (GX_ERR_INVALID), never silently drop the row."""
    li_keys = np.arange(1, 11, dtype=np.int64)
    price = np.full(10, 1.0); disc = np.zeros(10)
    ship = np.full(10, 9999, np.int32)
    # customer key 0
    cust, ordr, li = _mini_q3_tables(
        ctx, orc, li_keys, price, disc, ship,
        c_keys=np.arange(0, 100, dtype=np.int64))
    with pytest.raises(gx.GxError) as ei:
        ctx.q3(cust, ordr, li).run()
    assert ei.value.status == 3 and "sentinel" in str(ei.value)
    li.free(); ordr.free(); cust.free()
    # orders key 0
    cust, ordr, li = _mini_q3_tables(
        ctx, orc, li_keys, price, disc, ship,
        o_keys=np.arange(0, 100, dtype=np.int64))
    with pytest.raises(gx.GxError) as ei:
        ctx.q3(cust, ordr, li).run()
    assert ei.value.status == 3 and "sentinel" in str(ei.value)
    li.free(); ordr.free(); cust.free()


def test_hbm_budget_guard(ctx, orc, monkeypatch):
    """a build side beyond the HBM budget fails at sizing
    with the required-vs-available numbers, BEFORE any table allocation."""
    monkeypatch.setenv("GX_HBM_BUDGET_MB", "1")
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, 0.1)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, 0.1)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, 0.1)
    with pytest.raises(gx.GxError) as ei:
        ctx.q3(cust, ordr, li).run()
    assert ei.value.status == 5            # GX_ERR_OOM
    msg = str(ei.value)
    assert "HBM budget" in msg and "GB" in msg
    monkeypatch.delenv("GX_HBM_BUDGET_MB")
    # same tables run fine without the cap
    r = ctx.q3(cust, ordr, li).run().result()
    assert len(r["l_orderkey"]) > 0
    li.free(); ordr.free(); cust.free()


def test_partition_multi_bit_exact(ctx, orc):
    """Multi-column distribution keys (cdbhash.c:189-247 rotate-combine):
    GPU routing bit-exact vs the oracle for 1/2/3-key vectors, int8+int4
    type mixes, with and without NULL attributes."""
    rng = np.random.default_rng(41)
    for nkeys, types in ((1, [0]), (2, [0, 0]), (2, [0, 1]), (3, [0, 1, 0])):
        types = np.array(types, np.int32)
        vals = rng.integers(-2**62, 2**62, (20000, nkeys)).astype(np.int64)
        for k in range(nkeys):
            if types[k] == 1:
                vals[:, k] = rng.integers(-2**31, 2**31, 20000)
        nulls = (rng.random((20000, nkeys)) < 0.15).astype(np.uint8)
        for nsegs in (2, 3, 8, 64):
            got = ctx.partition_multi(vals, types, nsegs, isnull=nulls)
            want = orc.route_multi(vals, types, nsegs, isnull=nulls)
            np.testing.assert_array_equal(got, want)
            # NOT NULL fast path (isnull omitted)
            got_nn = ctx.partition_multi(vals, types, nsegs)
            want_nn = orc.route_multi(vals, types, nsegs)
            np.testing.assert_array_equal(got_nn, want_nn)
    # single-key multi == the established single-key entry
    keys = rng.integers(-2**62, 2**62, 5000).astype(np.int64)
    got = ctx.partition_multi(keys.reshape(-1, 1), np.zeros(1, np.int32), 8)
    np.testing.assert_array_equal(got, ctx.partition(keys, 8))


def test_extra_qual_lists(ctx, orc):
    """AND-ed qual lists per table (execScan.c:241 semantics over multiple
    quals): descriptor extra quals vs a numpy brute force of the same plan."""
    sf = 0.05
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    cut = gx.CUTOFF_19950315
    base = {"dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", cut),
            "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, ">", cut),
            # extra quals: priority in {1,2,3}, custkey even-ish bound,
            # lineitem shipdate upper bound AND orderkey range
            "mid_extra": [(3, ">=", 1), (3, "<=", 3), (1, "<", 40000)],
            "fact_extra": [(3, "<", cut + 900), (0, ">", 100)],
            "dim_extra": [(0, "<", 70000)]}
    got = ctx.q3_desc(base).run().result()

    c = orc.gen_customer(sf)
    o = orc.gen_orders(sf)
    w = orc.gen_lineitem(sf)
    cm = (c["c_mktsegment"] == 0) & (c["c_custkey"] < 70000)
    segok = c["c_custkey"][cm]
    om = ((o["o_orderdate"] < cut) & (o["o_shippriority"] >= 1) &
          (o["o_shippriority"] <= 3) & (o["o_custkey"] < 40000) &
          np.isin(o["o_custkey"], segok))
    lm = ((w["l_shipdate"] > cut) & (w["l_shipdate"] < cut + 900) &
          (w["l_orderkey"] > 100) &
          np.isin(w["l_orderkey"], o["o_orderkey"][om]))
    keys, counts = np.unique(w["l_orderkey"][lm], return_counts=True)
    np.testing.assert_array_equal(got["l_orderkey"], keys)
    np.testing.assert_array_equal(got["nitems"], counts)
    rev = {k: 0.0 for k in keys.tolist()}
    for k, p, dsc in zip(w["l_orderkey"][lm].tolist(),
                         w["l_extendedprice"][lm], w["l_discount"][lm]):
        rev[k] += p * (1.0 - dsc)
    np.testing.assert_allclose(got["revenue"],
                               np.array([rev[k] for k in keys.tolist()]),
                               rtol=1e-6)
    # no extras == classic plan (count the default path is untouched)
    plain = {k: v for k, v in base.items()
             if not k.endswith("_extra")}
    classic = ctx.q3(cust, ordr, li).run().result()
    via = ctx.q3_desc(plain).run().result()
    np.testing.assert_array_equal(via["l_orderkey"], classic["l_orderkey"])
    li.free(); ordr.free(); cust.free()


def test_extra_quals_with_visimap(ctx, orc):
    """Extra quals AND the AO visimap compose (the qual mask folds the
    visimap in at prepare)."""
    sf = 0.02
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    cut = gx.CUTOFF_19950315
    rng = np.random.default_rng(17)
    deleted = rng.random(li.nrows) < 0.10
    li.set_visimap(deleted)
    base = {"dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", cut),
            "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, ">", cut),
            "fact_extra": [(0, ">", 500)]}
    got = ctx.q3_desc(base).run().result()
    c = orc.gen_customer(sf)
    o = orc.gen_orders(sf)
    w = orc.gen_lineitem(sf)
    segok = c["c_custkey"][c["c_mktsegment"] == 0]
    om = (o["o_orderdate"] < cut) & np.isin(o["o_custkey"], segok)
    lm = ((w["l_shipdate"] > cut) & (w["l_orderkey"] > 500) & ~deleted &
          np.isin(w["l_orderkey"], o["o_orderkey"][om]))
    keys, counts = np.unique(w["l_orderkey"][lm], return_counts=True)
    np.testing.assert_array_equal(got["l_orderkey"], keys)
    np.testing.assert_array_equal(got["nitems"], counts)
    li.free(); ordr.free(); cust.free()


def test_null_join_keys_q3(ctx, orc):
    """NULL join keys through the pipeline.  Null-bearing
    key/filter columns (format-1 streams) are materialized at prepare with
    strict-NULL reject (nodeHash.c:2168-2181: a NULL key cannot pass the
    strict hash operator on either side of the inner join) and NULL-qual
    filtering (execScan.c:241).  Parity vs numpy brute force at ~10-15%
    NULLs on l_orderkey, o_custkey and o_orderdate."""
    rng = np.random.default_rng(55)
    nc, no, nl = 400, 1200, 4000
    c_keys = np.arange(1, nc + 1, dtype=np.int64)
    o_keys = np.arange(1, no + 1, dtype=np.int64)
    o_cust = rng.integers(1, nc + 1, no).astype(np.int64)
    o_cust_null = rng.random(no) < 0.12
    o_date = rng.integers(-3000, -1000, no).astype(np.int32)
    o_date_null = rng.random(no) < 0.08
    o_prio = rng.integers(0, 5, no).astype(np.int32)
    li_keys = rng.integers(1, no + 1, nl).astype(np.int64)
    li_null = rng.random(nl) < 0.15
    price = rng.uniform(1, 100, nl)
    disc = rng.integers(0, 11, nl) / 100.0
    ship = rng.integers(-2500, -500, nl).astype(np.int32)
    cut = -1753

    cust = ctx.bind([(orc.aocs_encode(c_keys), 8, nc),
                     (orc.aocs_encode(np.zeros(nc, np.int8)), 1, nc)])
    ordr = ctx.bind([(orc.aocs_encode(o_keys), 8, no),
                     (orc.aocs_encode_orig_nulls(o_cust, o_cust_null), 8, no, 1),
                     (orc.aocs_encode_orig_nulls(o_date, o_date_null), 4, no, 1),
                     (orc.aocs_encode(o_prio), 4, no)])
    li = ctx.bind([(orc.aocs_encode_rle_delta_nulls(li_keys, li_null), 8, nl, 1),
                   (orc.aocs_encode(price), 8, nl),
                   (orc.aocs_encode(disc), 8, nl),
                   (orc.aocs_encode(ship), 4, nl)])
    got = ctx.q3(cust, ordr, li).run().result()

    om = (~o_cust_null & ~o_date_null & (o_date < cut) &
          np.isin(o_cust, c_keys))
    lm = (~li_null & (ship > cut) & np.isin(li_keys, o_keys[om]))
    keys, counts = np.unique(li_keys[lm], return_counts=True)
    np.testing.assert_array_equal(got["l_orderkey"], keys)
    np.testing.assert_array_equal(got["nitems"], counts)
    rev = {k: 0.0 for k in keys.tolist()}
    for k, p, dsc in zip(li_keys[lm].tolist(), price[lm], disc[lm]):
        rev[k] += p * (1.0 - dsc)
    np.testing.assert_allclose(got["revenue"],
                               np.array([rev[k] for k in keys.tolist()]),
                               rtol=1e-6)
    li.free(); ordr.free(); cust.free()


def test_null_dim_key_strict_reject(ctx, orc):
    """NULL dim (customer) keys never enter the semijoin set."""
    rng = np.random.default_rng(56)
    nc = 300
    c_keys = np.arange(1, nc + 1, dtype=np.int64)
    c_null = rng.random(nc) < 0.2
    o_keys = np.arange(1, nc + 1, dtype=np.int64)
    o_cust = np.arange(1, nc + 1, dtype=np.int64)   # 1:1 with customers
    li_keys = np.arange(1, nc + 1, dtype=np.int64)
    cust = ctx.bind([(orc.aocs_encode_orig_nulls(c_keys, c_null), 8, nc, 1),
                     (orc.aocs_encode(np.zeros(nc, np.int8)), 1, nc)])
    ordr = ctx.bind([(orc.aocs_encode(o_keys), 8, nc),
                     (orc.aocs_encode(o_cust), 8, nc),
                     (orc.aocs_encode(np.full(nc, -9999, np.int32)), 4, nc),
                     (orc.aocs_encode(np.zeros(nc, np.int32)), 4, nc)])
    li = ctx.bind([(orc.aocs_encode(li_keys), 8, nc),
                   (orc.aocs_encode(np.full(nc, 10.0)), 8, nc),
                   (orc.aocs_encode(np.zeros(nc)), 8, nc),
                   (orc.aocs_encode(np.full(nc, 9999, np.int32)), 4, nc)])
    got = ctx.q3(cust, ordr, li).run().result()
    np.testing.assert_array_equal(got["l_orderkey"], c_keys[~c_null])
    li.free(); ordr.free(); cust.free()


def test_groupby_null_keys(ctx, orc):
    """NULLs-equal grouping (execGrouping.c:436-495): all NULL group keys
    form ONE group; SUM skips NULL inputs (strict transfn), COUNT(*) counts
    them.  GPU vs numpy on 15% NULL keys / 10% NULL values."""
    rng = np.random.default_rng(57)
    n = 50000
    keys = rng.integers(-50, 2000, n).astype(np.int64)
    knull = rng.random(n) < 0.15
    vals = rng.uniform(-5, 5, n)
    vnull = rng.random(n) < 0.10
    t = ctx.bind([(orc.aocs_encode_orig_nulls(keys, knull), 8, n, 1),
                  (orc.aocs_encode_orig_nulls(vals, vnull), 8, n, 1)])
    got = ctx.groupby(t, 0, 1)

    want_keys = np.unique(keys[~knull])
    body = got["key"][~got["key_is_null"]]
    np.testing.assert_array_equal(body, want_keys)
    for i, k in enumerate(want_keys):
        m = (~knull) & (keys == k)
        assert got["count"][i] == int(m.sum())
        np.testing.assert_allclose(got["sum"][i],
                                   vals[m & ~vnull].sum(), rtol=1e-9, atol=1e-9)
    # the NULL group is last
    assert got["key_is_null"][-1]
    assert got["count"][-1] == int(knull.sum())
    np.testing.assert_allclose(got["sum"][-1], vals[knull & ~vnull].sum(),
                               rtol=1e-9, atol=1e-9)
    t.free()


def test_groupby_not_null_columns(ctx, orc):
    """groupby on plain NOT NULL Orig columns (no NULL group at all)."""
    rng = np.random.default_rng(58)
    n = 10000
    keys = rng.integers(0, 500, n).astype(np.int64)
    vals = rng.uniform(0, 10, n)
    t = ctx.bind([(orc.aocs_encode(keys), 8, n),
                  (orc.aocs_encode(vals), 8, n)])
    got = ctx.groupby(t, 0, 1)
    want_keys = np.unique(keys)
    np.testing.assert_array_equal(got["key"], want_keys)
    assert not got["key_is_null"].any()
    for i, k in enumerate(want_keys):
        m = keys == k
        assert got["count"][i] == int(m.sum())
        np.testing.assert_allclose(got["sum"][i], vals[m].sum(), rtol=1e-9)
    t.free()


def test_rle_key_roles_materialized(ctx, orc):
    """RLE streams on NON-fact key roles (o_orderkey, o_custkey) now work
    via prepare-time materialization — results equal the plain-format run."""
    sf = 0.02
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    want = ctx.q3(cust, ordr, li).run().result()
    okeys = ordr.decode_column(0, np.int64)
    ocust = ordr.decode_column(1, np.int64)
    odate = ordr.decode_column(2, np.int32)
    oprio = ordr.decode_column(3, np.int32)
    no = len(okeys)
    ordr2 = ctx.bind([(orc.aocs_encode_rle_delta(okeys), 8, no, 1),
                      (orc.aocs_encode_rle(ocust), 8, no, 1),
                      (orc.aocs_encode(odate), 4, no),
                      (orc.aocs_encode(oprio), 4, no)])
    got = ctx.q3(cust, ordr2, li).run().result()
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-12)
    ordr2.free(); li.free(); ordr.free(); cust.free()


def test_q3_full_parity_sf25(ctx, orc):
    """the SF25 full-result comparison promoted from a
    DESIGN.md prose spot-check to a committed test — every group bit-exact
    on keys/dates/priority/counts, revenue within the stated tolerance,
    at 150M lineitem rows."""
    sf = 25.0
    orc.set_threads(0)            # all cores
    ng = _q3_parity_at(ctx, orc, sf)
    assert ng > 3_000_000


def _join_variety_tables(ctx, orc, rng, with_fk_nulls=False,
                         with_dim_nulls=False):
    nc, no, nl = 300, 1500, 5000
    c_keys = np.arange(1, nc + 1, dtype=np.int64)
    c_seg = (np.arange(nc) % 3).astype(np.int8)    # segment 0 = "BUILDING"
    o_keys = np.arange(1, no + 1, dtype=np.int64)
    o_cust = rng.integers(1, nc + 1, no).astype(np.int64)
    o_date = rng.integers(-3000, -1000, no).astype(np.int32)
    o_prio = rng.integers(0, 5, no).astype(np.int32)
    li_keys = rng.integers(1, no + 1, nl).astype(np.int64)
    price = rng.uniform(1, 100, nl)
    disc = rng.integers(0, 11, nl) / 100.0
    ship = rng.integers(-2500, -500, nl).astype(np.int32)
    fk_null = (rng.random(no) < 0.12) if with_fk_nulls else np.zeros(no, bool)
    dim_null = (rng.random(nc) < 0.1) if with_dim_nulls else np.zeros(nc, bool)
    if with_dim_nulls:
        cust = ctx.bind([(orc.aocs_encode_orig_nulls(c_keys, dim_null), 8, nc, 1),
                         (orc.aocs_encode(c_seg), 1, nc)])
    else:
        cust = ctx.bind([(orc.aocs_encode(c_keys), 8, nc),
                         (orc.aocs_encode(c_seg), 1, nc)])
    if with_fk_nulls:
        ordr = ctx.bind([(orc.aocs_encode(o_keys), 8, no),
                         (orc.aocs_encode_orig_nulls(o_cust, fk_null), 8, no, 1),
                         (orc.aocs_encode(o_date), 4, no),
                         (orc.aocs_encode(o_prio), 4, no)])
    else:
        ordr = ctx.bind([(orc.aocs_encode(o_keys), 8, no),
                         (orc.aocs_encode(o_cust), 8, no),
                         (orc.aocs_encode(o_date), 4, no),
                         (orc.aocs_encode(o_prio), 4, no)])
    li = ctx.bind([(orc.aocs_encode(li_keys), 8, nl),
                   (orc.aocs_encode(price), 8, nl),
                   (orc.aocs_encode(disc), 8, nl),
                   (orc.aocs_encode(ship), 4, nl)])
    data = dict(c_keys=c_keys, c_seg=c_seg, o_keys=o_keys, o_cust=o_cust,
                o_date=o_date, li_keys=li_keys, price=price, disc=disc,
                ship=ship, fk_null=fk_null, dim_null=dim_null)
    return cust, ordr, li, data


def _join_variety_desc(cust, ordr, li, dim_join):
    cut = -1753
    return {"dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", cut),
            "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, ">", cut), "dim_join": dim_join}


def _expect_groups(d, keep_orders_mask):
    cut = -1753
    lm = (d["ship"] > cut) & np.isin(d["li_keys"], d["o_keys"][keep_orders_mask])
    keys, counts = np.unique(d["li_keys"][lm], return_counts=True)
    return keys, counts


def test_anti_join_lasj(ctx, orc):
    """JOIN_LASJ (NOT EXISTS / anti) on the dim semijoin
    (nodeHashjoin.c:652-659): mid rows pass when the fk is NOT in the dim
    set; a NULL fk never matches, so it PASSES."""
    rng = np.random.default_rng(71)
    cust, ordr, li, d = _join_variety_tables(ctx, orc, rng, with_fk_nulls=True)
    got = ctx.q3_desc(_join_variety_desc(cust, ordr, li, "anti")).run().result()
    cut = -1753
    segok = d["c_keys"][d["c_seg"] == 0]
    in_set = np.isin(d["o_cust"], segok) & ~d["fk_null"]
    keep = (d["o_date"] < cut) & ~in_set     # NULL fk rows pass the anti join
    keys, counts = _expect_groups(d, keep)
    np.testing.assert_array_equal(got["l_orderkey"], keys)
    np.testing.assert_array_equal(got["nitems"], counts)
    li.free(); ordr.free(); cust.free()


def test_anti_join_lasj_notin(ctx, orc):
    """JOIN_LASJ_NOTIN (NOT IN): a NULL fk is REJECTED (NULL NOT IN (...)
    is unknown), and any NULL dim key passing the dim filter empties the
    whole result (nodeHashjoin.c:425,442)."""
    rng = np.random.default_rng(72)
    # case 1: NULL fks rejected, no dim NULLs
    cust, ordr, li, d = _join_variety_tables(ctx, orc, rng, with_fk_nulls=True)
    got = ctx.q3_desc(
        _join_variety_desc(cust, ordr, li, "anti_notin")).run().result()
    cut = -1753
    segok = d["c_keys"][d["c_seg"] == 0]
    in_set = np.isin(d["o_cust"], segok)
    keep = (d["o_date"] < cut) & ~in_set & ~d["fk_null"]
    keys, counts = _expect_groups(d, keep)
    np.testing.assert_array_equal(got["l_orderkey"], keys)
    np.testing.assert_array_equal(got["nitems"], counts)
    li.free(); ordr.free(); cust.free()

    # case 2: a NULL dim key passing the filter -> EMPTY result
    cust, ordr, li, d = _join_variety_tables(ctx, orc, rng,
                                             with_dim_nulls=True)
    q = ctx.q3_desc(_join_variety_desc(cust, ordr, li, "anti_notin")).run()
    got = q.result()
    # segment-0 rows include NULL keys with probability ~1 at this size
    has_null_in_filter = (d["dim_null"] & (d["c_seg"] == 0)).any()
    assert has_null_in_filter, "fixture must include a filtered NULL dim key"
    assert len(got["l_orderkey"]) == 0
    li.free(); ordr.free(); cust.free()


def test_semi_join_unchanged_by_dim_join_field(ctx, orc):
    """dim_join='semi' equals the classic default plan."""
    rng = np.random.default_rng(73)
    cust, ordr, li, d = _join_variety_tables(ctx, orc, rng)
    classic = ctx.q3(cust, ordr, li).run().result()
    semi = ctx.q3_desc(_join_variety_desc(cust, ordr, li, "semi")).run().result()
    np.testing.assert_array_equal(semi["l_orderkey"], classic["l_orderkey"])
    np.testing.assert_array_equal(semi["nitems"], classic["nitems"])
    li.free(); ordr.free(); cust.free()


def test_texteq_null_bearing_varlena(ctx, orc):
    """r2: the per-run texteq mask handles NULL-bearing varlena dim columns
    — a NULL mktsegment fails the qual (three-valued texteq), it no longer
    errors out."""
    rng = np.random.default_rng(81)
    nc = 500
    c_keys = np.arange(1, nc + 1, dtype=np.int64)
    segs = [b"BUILDING", b"AUTOMOBILE", b"MACHINERY"]
    seg_idx = rng.integers(0, 3, nc)
    nulls = rng.random(nc) < 0.2
    strings = [segs[int(i)] for i in seg_idx]
    cust = ctx.bind([
        (orc.aocs_encode(c_keys), 8, nc, 0),
        (orc.aocs_encode_varlena(strings, nulls=nulls), -1, nc, 1)])
    no = 1000
    o_keys = np.arange(1, no + 1, dtype=np.int64)
    o_cust = rng.integers(1, nc + 1, no).astype(np.int64)
    ordr = ctx.bind([(orc.aocs_encode(o_keys), 8, no),
                     (orc.aocs_encode(o_cust), 8, no),
                     (orc.aocs_encode(np.full(no, -9999, np.int32)), 4, no),
                     (orc.aocs_encode(np.zeros(no, np.int32)), 4, no)])
    nl = 3000
    li_keys = rng.integers(1, no + 1, nl).astype(np.int64)
    li = ctx.bind([(orc.aocs_encode(li_keys), 8, nl),
                   (orc.aocs_encode(np.full(nl, 5.0)), 8, nl),
                   (orc.aocs_encode(np.zeros(nl)), 8, nl),
                   (orc.aocs_encode(np.full(nl, 9999, np.int32)), 4, nl)])
    got = ctx.q3_desc({
        "dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", "BUILDING"),
        "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
        "mid_attr1_col": 2, "mid_attr2_col": 3,
        "mid_filter": (2, "<", -1753),
        "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
        "fact_filter": (3, ">", -1753)}).run().result()
    segok = c_keys[(seg_idx == 0) & ~nulls]
    om = np.isin(o_cust, segok)
    keys, counts = np.unique(li_keys[np.isin(li_keys, o_keys[om])],
                             return_counts=True)
    np.testing.assert_array_equal(got["l_orderkey"], keys)
    np.testing.assert_array_equal(got["nitems"], counts)
    li.free(); ordr.free(); cust.free()


def test_anti_join_forced_motion(ctx, orc):
    """Anti join through the FULL RCCL exchange branch (GX_FORCE_MOTION):
    the destination-bloom prefilter must stay OFF for anti joins — results
    equal the local anti run."""
    import os
    rng = np.random.default_rng(91)
    cust, ordr, li, d = _join_variety_tables(ctx, orc, rng, with_fk_nulls=True)
    want = ctx.q3_desc(_join_variety_desc(cust, ordr, li, "anti")).run().result()
    try:
        ctx.comm_init(ctx.comm_unique_id())
    except gx.GxError:
        pass    # communicator may already exist from an earlier test
    os.environ["GX_FORCE_MOTION"] = "1"
    try:
        got = ctx.q3_desc(
            _join_variety_desc(cust, ordr, li, "anti")).run().result()
    finally:
        del os.environ["GX_FORCE_MOTION"]
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-12)
    li.free(); ordr.free(); cust.free()


def test_groupby_rle_key_column(ctx, orc):
    """gx_groupby over a Dense/RLE (no-null) key column — decoded at entry,
    same groups as the plain-format bind."""
    rng = np.random.default_rng(92)
    n = 30000
    keys = np.repeat(rng.integers(1, 900, 600).astype(np.int64), 50)
    vals = rng.uniform(0, 4, n)
    t_plain = ctx.bind([(orc.aocs_encode(keys), 8, n),
                        (orc.aocs_encode(vals), 8, n)])
    t_rle = ctx.bind([(orc.aocs_encode_rle(keys), 8, n, 1),
                      (orc.aocs_encode(vals), 8, n)])
    a = ctx.groupby(t_plain, 0, 1)
    b = ctx.groupby(t_rle, 0, 1)
    np.testing.assert_array_equal(a["key"], b["key"])
    np.testing.assert_array_equal(a["count"], b["count"])
    np.testing.assert_allclose(a["sum"], b["sum"], rtol=1e-9)
    t_rle.free(); t_plain.free()


def test_extra_quals_on_materialized_table(ctx, orc):
    """Extra quals (on plain columns) compose with a NULL-bearing
    materialized key column on the same table: the hidden mask carries
    both the qual failures and the strict-NULL rejects."""
    rng = np.random.default_rng(93)
    nl = 4000
    li_keys = rng.integers(1, 101, nl).astype(np.int64)
    li_null = rng.random(nl) < 0.2
    price = np.full(nl, 2.0); disc = np.zeros(nl)
    ship = rng.integers(-3000, 3000, nl).astype(np.int32)
    cust, ordr, li = _mini_q3_tables(
        ctx, orc, li_keys, price, disc, ship)
    li.free()
    li = ctx.bind([(orc.aocs_encode_orig_nulls(li_keys, li_null), 8, nl, 1),
                   (orc.aocs_encode(price), 8, nl),
                   (orc.aocs_encode(disc), 8, nl),
                   (orc.aocs_encode(ship), 4, nl)])
    cut = 0
    got = ctx.q3_desc({
        "dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
        "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
        "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", cut),
        "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
        "fact_filter": (3, ">", -2000),
        "fact_extra": [(3, "<", 2000), (0, ">", 10)]}).run().result()
    lm = (~li_null & (ship > -2000) & (ship < 2000) & (li_keys > 10))
    keys, counts = np.unique(li_keys[lm], return_counts=True)
    np.testing.assert_array_equal(got["l_orderkey"], keys)
    np.testing.assert_array_equal(got["nitems"], counts)
    li.free(); ordr.free(); cust.free()


def test_q3_desc_fuzz(ctx, orc):
    """Randomized plan fuzz over the widened descriptor surface: random
    comparison ops/literals, 0-3 extra quals per table, all three dim-join
    types, optional NULL-bearing key columns and visimaps — every plan
    checked against an independent numpy evaluation.  Seeded: failures
    reproduce."""
    rng = np.random.default_rng(1234)
    OPS = ["<", ">", "==", "!=", "<=", ">="]

    def np_cmp(v, op, lit):
        return {"<": v < lit, ">": v > lit, "==": v == lit, "!=": v != lit,
                "<=": v <= lit, ">=": v >= lit}[op]

    for trial in range(12):
        nc = int(rng.integers(50, 400))
        no = int(rng.integers(200, 1500))
        nl = int(rng.integers(500, 5000))
        c_keys = np.arange(1, nc + 1, dtype=np.int64)
        c_seg = rng.integers(0, 4, nc).astype(np.int8)
        o_keys = np.arange(1, no + 1, dtype=np.int64)
        o_cust = rng.integers(1, nc + 1, no).astype(np.int64)
        o_date = rng.integers(-400, 400, no).astype(np.int32)
        o_prio = rng.integers(0, 5, no).astype(np.int32)
        li_keys = rng.integers(1, no + 1, nl).astype(np.int64)
        price = rng.uniform(1, 50, nl)
        disc = rng.integers(0, 11, nl) / 100.0
        ship = rng.integers(-400, 400, nl).astype(np.int32)

        with_li_nulls = bool(rng.random() < 0.4)
        li_null = (rng.random(nl) < 0.15) if with_li_nulls else np.zeros(nl, bool)
        with_vmap = bool(rng.random() < 0.3)
        deleted = (rng.random(nl) < 0.1) if with_vmap else np.zeros(nl, bool)

        dim_join = ["semi", "anti", "anti_notin"][int(rng.integers(0, 3))]

        cust = ctx.bind([(orc.aocs_encode(c_keys), 8, nc),
                         (orc.aocs_encode(c_seg), 1, nc)])
        ordr = ctx.bind([(orc.aocs_encode(o_keys), 8, no),
                         (orc.aocs_encode(o_cust), 8, no),
                         (orc.aocs_encode(o_date), 4, no),
                         (orc.aocs_encode(o_prio), 4, no)])
        if with_li_nulls:
            li_key_stream = (orc.aocs_encode_orig_nulls(li_keys, li_null), 8, nl, 1)
        else:
            li_key_stream = (orc.aocs_encode(li_keys), 8, nl)
        li = ctx.bind([li_key_stream,
                       (orc.aocs_encode(price), 8, nl),
                       (orc.aocs_encode(disc), 8, nl),
                       (orc.aocs_encode(ship), 4, nl)])
        if with_vmap:
            li.set_visimap(deleted)

        dop = OPS[int(rng.integers(0, 6))]
        dlit = int(rng.integers(0, 4))
        mop = OPS[int(rng.integers(0, 6))]
        mlit = int(rng.integers(-300, 300))
        fop = OPS[int(rng.integers(0, 6))]
        flit = int(rng.integers(-300, 300))
        n_mid_x = int(rng.integers(0, 3))
        n_fact_x = int(rng.integers(0, 3))
        mid_x = [(3, OPS[int(rng.integers(0, 6))], int(rng.integers(0, 5)))
                 for _ in range(n_mid_x)]
        fact_x = [(3, OPS[int(rng.integers(0, 6))], int(rng.integers(-300, 300)))
                  for _ in range(n_fact_x)]

        got = ctx.q3_desc({
            "dim": cust, "dim_key_col": 0, "dim_filter": (1, dop, dlit),
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3,
            "mid_filter": (2, mop, mlit),
            "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, fop, flit),
            "mid_extra": mid_x, "fact_extra": fact_x,
            "dim_join": dim_join}).run().result()

        # independent evaluation
        segok = c_keys[np_cmp(c_seg, dop, dlit)]
        in_set = np.isin(o_cust, segok)
        om = np_cmp(o_date, mop, mlit)
        for col, op, lit in mid_x:
            om &= np_cmp(o_prio, op, lit)
        if dim_join == "semi":
            om &= in_set
        else:           # anti / anti_notin: NOT-NULL fks, plain complement
            om &= ~in_set
        lm = np_cmp(ship, fop, flit) & ~li_null & ~deleted
        for col, op, lit in fact_x:
            lm &= np_cmp(ship, op, lit)
        lm &= np.isin(li_keys, o_keys[om])
        keys, counts = np.unique(li_keys[lm], return_counts=True)
        np.testing.assert_array_equal(got["l_orderkey"], keys,
                                      err_msg=f"trial {trial}")
        np.testing.assert_array_equal(got["nitems"], counts,
                                      err_msg=f"trial {trial}")
        rev = {k: 0.0 for k in keys.tolist()}
        for k, p, dsc in zip(li_keys[lm].tolist(), price[lm], disc[lm]):
            rev[k] += p * (1.0 - dsc)
        np.testing.assert_allclose(
            got["revenue"], np.array([rev[k] for k in keys.tolist()]),
            rtol=1e-6, err_msg=f"trial {trial}")
        li.free(); ordr.free(); cust.free()


def test_left_outer_fact_join(ctx, orc):
    """LEFT OUTER fact⋈mid (HJ_FILL_OUTER): fact rows passing their WHERE
    quals but matching no qualifying order form groups with NULL mid attrs;
    NULL fact keys all land in ONE NULL-key group, returned last."""
    rng = np.random.default_rng(101)
    nc, no, nl = 200, 600, 4000
    c_keys = np.arange(1, nc + 1, dtype=np.int64)
    c_seg = (np.arange(nc) % 3).astype(np.int8)
    o_keys = np.arange(1, no + 1, dtype=np.int64)
    o_cust = rng.integers(1, nc + 1, no).astype(np.int64)
    o_date = rng.integers(-400, 400, no).astype(np.int32)
    o_prio = rng.integers(0, 5, no).astype(np.int32)
    # lineitem keys: some matching, some beyond the orders domain, some NULL
    li_keys = rng.integers(1, no + 300, nl).astype(np.int64)
    li_null = rng.random(nl) < 0.1
    price = rng.uniform(1, 20, nl)
    disc = rng.integers(0, 11, nl) / 100.0
    ship = rng.integers(-400, 400, nl).astype(np.int32)
    cut = 0
    cust = ctx.bind([(orc.aocs_encode(c_keys), 8, nc),
                     (orc.aocs_encode(c_seg), 1, nc)])
    ordr = ctx.bind([(orc.aocs_encode(o_keys), 8, no),
                     (orc.aocs_encode(o_cust), 8, no),
                     (orc.aocs_encode(o_date), 4, no),
                     (orc.aocs_encode(o_prio), 4, no)])
    li = ctx.bind([(orc.aocs_encode_orig_nulls(li_keys, li_null), 8, nl, 1),
                   (orc.aocs_encode(price), 8, nl),
                   (orc.aocs_encode(disc), 8, nl),
                   (orc.aocs_encode(ship), 4, nl)])
    got = ctx.q3_desc({
        "dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
        "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
        "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", cut),
        "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
        "fact_filter": (3, ">", cut),
        "fact_join": "left_outer"}).run().result()

    segok = c_keys[c_seg == 0]
    om = (o_date < cut) & np.isin(o_cust, segok)
    qual_keys = o_keys[om]
    lm = (ship > cut)                      # WHERE on the left table
    matched = lm & ~li_null & np.isin(li_keys, qual_keys)
    unmatched = lm & ~li_null & ~np.isin(li_keys, qual_keys)
    nullkey = lm & li_null

    mk, mc = np.unique(li_keys[matched], return_counts=True)
    uk, uc = np.unique(li_keys[unmatched], return_counts=True)

    m_got = got["l_orderkey"][~got["attrs_null"]]
    np.testing.assert_array_equal(np.sort(m_got), mk)
    u_mask = got["attrs_null"] & ~got["key_is_null"]
    np.testing.assert_array_equal(np.sort(got["l_orderkey"][u_mask]), uk)
    # counts per group
    order = np.argsort(got["l_orderkey"][u_mask])
    np.testing.assert_array_equal(got["nitems"][u_mask][order], uc)
    assert (got["o_orderdate"][got["attrs_null"]] == 0).all()
    # NULL-key group: one, last, counts all null-key rows
    nk = got["key_is_null"]
    if nullkey.any():
        assert nk.sum() == 1 and nk[-1]
        assert got["nitems"][nk][0] == int(nullkey.sum())
        np.testing.assert_allclose(
            got["revenue"][nk][0],
            (price[nullkey] * (1 - disc[nullkey])).sum(), rtol=1e-9)
    else:
        assert nk.sum() == 0
    # revenue parity on unmatched groups
    rev = {k: 0.0 for k in uk.tolist()}
    for k, p, dsc in zip(li_keys[unmatched].tolist(),
                         price[unmatched], disc[unmatched]):
        rev[k] += p * (1.0 - dsc)
    np.testing.assert_allclose(
        got["revenue"][u_mask][order],
        np.array([rev[k] for k in uk.tolist()]), rtol=1e-6)
    li.free(); ordr.free(); cust.free()


def test_left_outer_inner_equivalence(ctx, orc):
    """With every fact key matched and no NULLs, left-outer equals inner."""
    sf = 0.02
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    cut = gx.CUTOFF_19950315
    base = {"dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", cut),
            "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, ">", cut)}
    inner = ctx.q3_desc(base).run().result()
    outer = ctx.q3_desc(dict(base, fact_join="left_outer")).run().result()
    # inner groups = outer groups with attrs present
    m = ~outer["attrs_null"]
    np.testing.assert_array_equal(outer["l_orderkey"][m], inner["l_orderkey"])
    np.testing.assert_array_equal(outer["o_orderdate"][m], inner["o_orderdate"])
    np.testing.assert_array_equal(outer["nitems"][m], inner["nitems"])
    np.testing.assert_allclose(outer["revenue"][m], inner["revenue"], rtol=1e-9)
    # every unmatched group comes from a real lineitem key outside the
    # qualifying set and carries NULL attrs
    assert (outer["o_orderdate"][~m] == 0).all()
    li.free(); ordr.free(); cust.free()


def test_left_outer_forced_motion(ctx, orc):
    """LEFT OUTER through the FULL RCCL exchange branch equals the local
    outer run (unmatched side is always the local fact shard)."""
    import os
    sf = 0.05
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    cut = gx.CUTOFF_19950315
    base = {"dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", cut),
            "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, ">", cut), "fact_join": "left_outer"}
    want = ctx.q3_desc(base).run().result()
    try:
        ctx.comm_init(ctx.comm_unique_id())
    except gx.GxError:
        pass
    os.environ["GX_FORCE_MOTION"] = "1"
    try:
        got = ctx.q3_desc(base).run().result()
    finally:
        del os.environ["GX_FORCE_MOTION"]
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    np.testing.assert_array_equal(got["attrs_null"], want["attrs_null"])
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-12)
    li.free(); ordr.free(); cust.free()


def test_left_outer_numeric(ctx, orc):
    """numeric(15,2) + LEFT OUTER: unmatched groups' revenue numerators are
    exact integers too."""
    sf = 0.02
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li_f = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    li_n = ctx.tpch_gen(gx.TPCH_LINEITEM_NUMERIC, sf)
    cut = gx.CUTOFF_19950315
    base = {"dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", cut),
            "fact": li_f, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, ">", cut), "fact_join": "left_outer"}
    f64 = ctx.q3_desc(base).run().result()
    qn = ctx.q3_desc(dict(base, fact=li_n))
    ctx._chk(ctx._lib.gx_q3_set_numeric(qn._q, 1))
    num = qn.run().result()
    np.testing.assert_array_equal(num["l_orderkey"], f64["l_orderkey"])
    np.testing.assert_array_equal(num["nitems"], f64["nitems"])
    np.testing.assert_array_equal(num["attrs_null"], f64["attrs_null"])
    # exact integer numerators, scale 1e-4
    np.testing.assert_allclose(num["revenue_num"] * 1e-4, f64["revenue"],
                               rtol=1e-9)
    li_n.free(); li_f.free(); ordr.free(); cust.free()


def test_anti_notin_nulls_failing_filter_not_empty(ctx, orc):
    """LASJ_NOTIN: a NULL dim key that FAILS the dim filter never reaches
    the hash build, so it must NOT empty the result (nodeHashjoin.c:442
    counts nulls seen during the build, i.e., after the scan quals)."""
    rng = np.random.default_rng(111)
    nc = 300
    c_keys = np.arange(1, nc + 1, dtype=np.int64)
    c_seg = (np.arange(nc) % 3).astype(np.int8)
    # NULL keys only on rows whose segment != 0 (they fail the filter)
    dim_null = (rng.random(nc) < 0.3) & (c_seg != 0)
    assert dim_null.any()
    cust = ctx.bind([(orc.aocs_encode_orig_nulls(c_keys, dim_null), 8, nc, 1),
                     (orc.aocs_encode(c_seg), 1, nc)])
    no, nl = 900, 2500
    o_keys = np.arange(1, no + 1, dtype=np.int64)
    o_cust = rng.integers(1, nc + 1, no).astype(np.int64)
    ordr = ctx.bind([(orc.aocs_encode(o_keys), 8, no),
                     (orc.aocs_encode(o_cust), 8, no),
                     (orc.aocs_encode(np.full(no, -9999, np.int32)), 4, no),
                     (orc.aocs_encode(np.zeros(no, np.int32)), 4, no)])
    li_keys = rng.integers(1, no + 1, nl).astype(np.int64)
    li = ctx.bind([(orc.aocs_encode(li_keys), 8, nl),
                   (orc.aocs_encode(np.full(nl, 3.0)), 8, nl),
                   (orc.aocs_encode(np.zeros(nl)), 8, nl),
                   (orc.aocs_encode(np.full(nl, 9999, np.int32)), 4, nl)])
    got = ctx.q3_desc({
        "dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
        "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
        "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", 0),
        "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
        "fact_filter": (3, ">", 0),
        "dim_join": "anti_notin"}).run().result()
    # NOT IN over the non-null segment-0 keys (none of which are NULL here)
    segok = c_keys[(c_seg == 0) & ~dim_null]
    om = ~np.isin(o_cust, segok)
    keys = np.unique(li_keys[np.isin(li_keys, o_keys[om])])
    assert len(got["l_orderkey"]) > 0          # NOT emptied
    np.testing.assert_array_equal(got["l_orderkey"], keys)
    li.free(); ordr.free(); cust.free()


def test_anti_lasj_with_null_dim_keys(ctx, orc):
    """Plain LASJ with NULL dim keys: strict build-side reject means NULL
    keys never enter the set; the anti join is simply the complement of
    the non-null filtered keys (no emptiness rule for LASJ)."""
    rng = np.random.default_rng(112)
    nc = 250
    c_keys = np.arange(1, nc + 1, dtype=np.int64)
    c_seg = (np.arange(nc) % 2).astype(np.int8)
    dim_null = rng.random(nc) < 0.25
    cust = ctx.bind([(orc.aocs_encode_orig_nulls(c_keys, dim_null), 8, nc, 1),
                     (orc.aocs_encode(c_seg), 1, nc)])
    no, nl = 800, 2000
    o_keys = np.arange(1, no + 1, dtype=np.int64)
    o_cust = rng.integers(1, nc + 1, no).astype(np.int64)
    ordr = ctx.bind([(orc.aocs_encode(o_keys), 8, no),
                     (orc.aocs_encode(o_cust), 8, no),
                     (orc.aocs_encode(np.full(no, -9999, np.int32)), 4, no),
                     (orc.aocs_encode(np.zeros(no, np.int32)), 4, no)])
    li_keys = rng.integers(1, no + 1, nl).astype(np.int64)
    li = ctx.bind([(orc.aocs_encode(li_keys), 8, nl),
                   (orc.aocs_encode(np.full(nl, 3.0)), 8, nl),
                   (orc.aocs_encode(np.zeros(nl)), 8, nl),
                   (orc.aocs_encode(np.full(nl, 9999, np.int32)), 4, nl)])
    got = ctx.q3_desc({
        "dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
        "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
        "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", 0),
        "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
        "fact_filter": (3, ">", 0),
        "dim_join": "anti"}).run().result()
    segok = c_keys[(c_seg == 0) & ~dim_null]   # strict: nulls never build
    om = ~np.isin(o_cust, segok)
    keys = np.unique(li_keys[np.isin(li_keys, o_keys[om])])
    np.testing.assert_array_equal(got["l_orderkey"], keys)
    li.free(); ordr.free(); cust.free()


def test_left_outer_sf100_conservation(ctx, orc):
    """Full-size LEFT OUTER: at SF100 every shipdate-passing lineitem row
    lands in exactly ONE group (matched or NULL-attr unmatched), so
    sum(nitems) equals the standalone filter count; matched items equal
    the inner join's probe hits."""
    sf = 100.0
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    cut = gx.CUTOFF_19950315
    nfilt, _ = li.scan_filter(3, ">", cut)
    qi = ctx.q3(cust, ordr, li).run()
    inner_hits = qi.stats()["probe_hits"]
    base = {"dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", cut),
            "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, ">", cut), "fact_join": "left_outer"}
    q = ctx.q3_desc(base).run()
    r = q.result()
    assert int(r["nitems"].sum()) == int(nfilt)
    m = ~r["attrs_null"]
    assert int(r["nitems"][m].sum()) == int(inner_hits)
    assert not r["key_is_null"].any()         # synthetic keys are NOT NULL
    # group keys unique across matched+unmatched
    assert len(np.unique(r["l_orderkey"])) == len(r["l_orderkey"])
    q.free(); li.free(); ordr.free(); cust.free()


def test_topn_left_outer_nulls_last(ctx, orc):
    """gx_q3_topn over an outer plan: ORDER BY revenue DESC, o_orderdate
    with NULL dates ranking LAST on revenue ties (PG NULLS LAST)."""
    rng = np.random.default_rng(121)
    cust, ordr, li, d = _join_variety_tables(ctx, orc, rng)
    base = _join_variety_desc(cust, ordr, li, "semi")
    base["fact_join"] = "left_outer"
    q = ctx.q3_desc(base).run()
    r = q.result()
    top = q.topn(10)
    # host reference: sort all groups by (rev desc, eff-date asc)
    eff = np.where(r["attrs_null"], np.iinfo(np.int32).max, r["o_orderdate"])
    order = np.lexsort((eff, -r["revenue"]))
    want = order[:len(top["l_orderkey"])]
    np.testing.assert_allclose(top["revenue"], r["revenue"][want], rtol=1e-12)
    np.testing.assert_array_equal(top["attrs_null"], r["attrs_null"][want])
    li.free(); ordr.free(); cust.free()


def test_left_outer_rle_fact_key(ctx, orc):
    """Outer + RLE fact key: the fused-RLE scan is inner-only, so the key
    column materializes at prepare — results equal the plain-format outer
    run."""
    sf = 0.02
    cust = ctx.tpch_gen(gx.TPCH_CUSTOMER, sf)
    ordr = ctx.tpch_gen(gx.TPCH_ORDERS, sf)
    li = ctx.tpch_gen(gx.TPCH_LINEITEM, sf)
    li_rle = ctx.tpch_gen(gx.TPCH_LINEITEM_RLEKEY, sf)
    cut = gx.CUTOFF_19950315
    base = {"dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3, "mid_filter": (2, "<", cut),
            "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, ">", cut), "fact_join": "left_outer"}
    want = ctx.q3_desc(base).run().result()
    got = ctx.q3_desc(dict(base, fact=li_rle)).run().result()
    np.testing.assert_array_equal(got["l_orderkey"], want["l_orderkey"])
    np.testing.assert_array_equal(got["nitems"], want["nitems"])
    np.testing.assert_array_equal(got["attrs_null"], want["attrs_null"])
    np.testing.assert_allclose(got["revenue"], want["revenue"], rtol=1e-12)
    li_rle.free(); li.free(); ordr.free(); cust.free()


def test_partition_multi_boundaries(ctx, orc):
    """Boundary attributes through the GPU rotate-combine chain."""
    cases = [
        (np.array([[0, 0]], np.int64), [0, 1]),
        (np.array([[2**63 - 1, -2**63]], np.int64), [0, 0]),
        (np.array([[-2**31, 2**31 - 1]], np.int64), [1, 1]),
        (np.array([[-1, 1, 0]], np.int64), [0, 1, 0]),
    ]
    for vals, types in cases:
        t = np.array(types, np.int32)
        for nsegs in (2, 8, 64):
            got = ctx.partition_multi(vals, t, nsegs)
            want = orc.route_multi(vals, t, nsegs)
            np.testing.assert_array_equal(got, want)


def test_q3_desc_fuzz_outer(ctx, orc):
    """Plan fuzz over the fact-join axis: random inner/left_outer plans
    with NULL-bearing fact keys, checked against a numpy outer model
    (matched groups + NULL-attr unmatched groups + one NULL-key group)."""
    rng = np.random.default_rng(4321)
    OPS = ["<", ">", "==", "!=", "<=", ">="]

    def np_cmp(v, op, lit):
        return {"<": v < lit, ">": v > lit, "==": v == lit, "!=": v != lit,
                "<=": v <= lit, ">=": v >= lit}[op]

    for trial in range(8):
        nc = int(rng.integers(50, 300))
        no = int(rng.integers(200, 1000))
        nl = int(rng.integers(500, 4000))
        c_keys = np.arange(1, nc + 1, dtype=np.int64)
        c_seg = rng.integers(0, 4, nc).astype(np.int8)
        o_keys = np.arange(1, no + 1, dtype=np.int64)
        o_cust = rng.integers(1, nc + 1, no).astype(np.int64)
        o_date = rng.integers(-400, 400, no).astype(np.int32)
        o_prio = rng.integers(0, 5, no).astype(np.int32)
        li_keys = rng.integers(1, no + 200, nl).astype(np.int64)
        with_nulls = bool(rng.random() < 0.5)
        li_null = (rng.random(nl) < 0.12) if with_nulls else np.zeros(nl, bool)
        price = rng.uniform(1, 30, nl)
        disc = rng.integers(0, 11, nl) / 100.0
        ship = rng.integers(-400, 400, nl).astype(np.int32)
        fop = OPS[int(rng.integers(0, 6))]
        flit = int(rng.integers(-300, 300))
        mop = OPS[int(rng.integers(0, 6))]
        mlit = int(rng.integers(-300, 300))

        cust = ctx.bind([(orc.aocs_encode(c_keys), 8, nc),
                         (orc.aocs_encode(c_seg), 1, nc)])
        ordr = ctx.bind([(orc.aocs_encode(o_keys), 8, no),
                         (orc.aocs_encode(o_cust), 8, no),
                         (orc.aocs_encode(o_date), 4, no),
                         (orc.aocs_encode(o_prio), 4, no)])
        if with_nulls:
            key_stream = (orc.aocs_encode_orig_nulls(li_keys, li_null), 8, nl, 1)
        else:
            key_stream = (orc.aocs_encode(li_keys), 8, nl)
        li = ctx.bind([key_stream,
                       (orc.aocs_encode(price), 8, nl),
                       (orc.aocs_encode(disc), 8, nl),
                       (orc.aocs_encode(ship), 4, nl)])
        got = ctx.q3_desc({
            "dim": cust, "dim_key_col": 0, "dim_filter": (1, "==", 0),
            "mid": ordr, "mid_key_col": 0, "mid_fk_col": 1,
            "mid_attr1_col": 2, "mid_attr2_col": 3,
            "mid_filter": (2, mop, mlit),
            "fact": li, "fact_key_col": 0, "fact_a_col": 1, "fact_b_col": 2,
            "fact_filter": (3, fop, flit),
            "fact_join": "left_outer"}).run().result()

        segok = c_keys[c_seg == 0]
        om = np_cmp(o_date, mop, mlit) & np.isin(o_cust, segok)
        qual_keys = o_keys[om]
        lm = np_cmp(ship, fop, flit)
        matched = lm & ~li_null & np.isin(li_keys, qual_keys)
        unmatched = lm & ~li_null & ~np.isin(li_keys, qual_keys)
        nullrows = lm & li_null

        mk, mc2 = np.unique(li_keys[matched], return_counts=True)
        uk, uc2 = np.unique(li_keys[unmatched], return_counts=True)
        gm = ~got["attrs_null"]
        gu = got["attrs_null"] & ~got["key_is_null"]
        np.testing.assert_array_equal(np.sort(got["l_orderkey"][gm]), mk,
                                      err_msg=f"trial {trial} matched keys")
        np.testing.assert_array_equal(np.sort(got["l_orderkey"][gu]), uk,
                                      err_msg=f"trial {trial} unmatched keys")
        order = np.argsort(got["l_orderkey"][gu])
        np.testing.assert_array_equal(got["nitems"][gu][order], uc2,
                                      err_msg=f"trial {trial} unmatched counts")
        nk = got["key_is_null"]
        if nullrows.any():
            assert nk.sum() == 1 and int(got["nitems"][nk][0]) == int(nullrows.sum()), \
                f"trial {trial} null group"
        else:
            assert nk.sum() == 0, f"trial {trial} spurious null group"
        assert int(got["nitems"].sum()) == int(matched.sum() + unmatched.sum()
                                               + nullrows.sum())
        li.free(); ordr.free(); cust.free()
